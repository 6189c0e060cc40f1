// Common device helpers for the r2d2_amd gfx950 kernels.
// Wavefront width on CDNA4 is 64 (not 32) — every cross-lane idiom below is
// 64-wide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

// ---------------------------------------------------------------------------
// R2D2 value rescaling h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x and its inverse
// (golden: r2d2_amd/ops/functional.py; reference semantics worker.py:383-390)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float value_rescale(float x, float eps) {
    float s = (x > 0.f) - (x < 0.f);
    return s * (sqrtf(fabsf(x) + 1.f) - 1.f) + eps * x;
}

__device__ __forceinline__ float inv_value_rescale(float x, float eps) {
    float s = (x > 0.f) - (x < 0.f);
    float t = (sqrtf(1.f + 4.f * eps * (fabsf(x) + 1.f + eps)) - 1.f) / (2.f * eps);
    return s * (t * t - 1.f);
}

// ---------------------------------------------------------------------------
// Wave-wide reductions (64 lanes)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
    return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off));
    return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_allreduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
    return v;
}

__device__ __forceinline__ float wave_allreduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
    return v;
}

// ---------------------------------------------------------------------------
// MFMA column-fragment gather via gfx950 hardware transpose-reads.
//   returns a[e] = s[(row0 + e) * LDE + col0 + (lane & 15)], e = 0..7
// as TWO ds_read_b64_tr_b16 instead of 8 scalar LDS reads (the wgrad
// outer-product transpose).  Each 16-lane group's lane i supplies the
// address of the 4-element piece (row row0 + i/4, cols col0 + 4*(i%4));
// the instruction redistributes so lane c receives column c of the
// group's 4x16 block — semantics verified empirically on MI355X
// (tools/tr16_probe.hip -> gpurun_out/tr16_semantics.log mode 1).
// Requires col0 % 4 == 0 and the 16-column span in-bounds.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) __bf16 cmn_bf16x4;
typedef __attribute__((ext_vector_type(8))) __bf16 cmn_bf16x8;

template <int LDE>
__device__ __forceinline__ cmn_bf16x8 lds_col_frag8(
    const __hip_bfloat16* s, int row0, int col0, int lane) {
    int i15 = lane & 15;
    auto* p = (__attribute__((address_space(3))) cmn_bf16x4*)
        (s + (row0 + (i15 >> 2)) * LDE + col0 + 4 * (i15 & 3));
    union {
        struct { cmn_bf16x4 lo, hi; } p2;
        cmn_bf16x8 v;
    } u;
    u.p2.lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);      // rows 0..3
    u.p2.hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p + LDE); // rows 4..7
    return u.v;
}

// bf16 <-> f32 helpers ------------------------------------------------------
__device__ __forceinline__ float bf2f(__hip_bfloat16 v) {
    return __bfloat162float(v);
}
__device__ __forceinline__ __hip_bfloat16 f2bf(float v) {
    return __float2bfloat16(v);
}

#define HIP_CHECK(expr)                                                        \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess) {                                                \
            printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, \
                   __LINE__);                                                  \
        }                                                                      \
    } while (0)
