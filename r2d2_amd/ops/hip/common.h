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
