// MFMA GEMM kernels for the R2D2 network's dense layers (gfx950).
//
// C(M,N) = A(M,K) @ W^T + bias, optional ReLU.  Weights are prepacked
// row-major as Wt(N,K) so both A and B fragments load 8 CONTIGUOUS bf16
// (16 B) per lane — the mfma_f32_16x16x32_bf16 operand layout puts 8
// consecutive k-elements in each lane (A: row = lane&15; B: col = lane&15;
// k = (lane>>4)*8 + j).  Replaces hipBLASLt/eager Linear for the encoder FC
// (3136->512), the dueling heads, and the LSTM gate GEMMs.
//
// Structure: 4-wave workgroups, 64x64 block tile, each wave a 32x32 tile
// (2x2 fragments, A/B frags shared across the wave's row/col pair), K-major
// loop.  Weights are L2-resident (<= 3.2 MB); correctness-first, tuned via
// rocprof (see profiles/).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

union bf8u {
    bf16x8 v;
    uint4 u;
    __bf16 e[8];
};

__device__ __forceinline__ bf16x8 load_bf16x8(const __hip_bfloat16* p) {
    bf8u r;
    r.u = *reinterpret_cast<const uint4*>(p);
    return r.v;
}

__device__ __forceinline__ bf16x8 zero_bf16x8() {
    bf8u r;
    r.u = uint4{0, 0, 0, 0};
    return r.v;
}

// ---------------------------------------------------------------------------
// gemm_bias_act: out(M,N) = act(A(M,K) @ Wt(N,K)^T + bias)
//   OUT_F32: write f32 (head outputs feeding the loss kernel) else bf16.
//   ACT: 0 none, 1 relu.
// ---------------------------------------------------------------------------
template <int ACT, bool HAS_BIAS, bool OUT_F32>
__global__ __launch_bounds__(256) void gemm_bias_act_kernel(
    const __hip_bfloat16* __restrict__ A,   // (M, K)
    const __hip_bfloat16* __restrict__ Wt,  // (N, K)
    const float* __restrict__ bias,         // (N,)
    void* __restrict__ out,                 // (M, N) bf16 or f32
    int M, int N, int K) {
    // block tile 64x64: wave w covers rows [wr*32, +32), cols [wc*32, +32)
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave >> 1, wc = wave & 1;
    long row0 = (long)blockIdx.x * 64 + wr * 32;
    long col0 = (long)blockIdx.y * 64 + wc * 32;

    int frow = lane & 15;            // fragment row (A) / col (B)
    int kseg = (lane >> 4) * 8;      // k offset of this lane's 8 elements

    f32x4 acc[2][2] = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        bf16x8 a[2], b[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            long r = row0 + i * 16 + frow;
            a[i] = (r < M) ? load_bf16x8(A + r * K + k0 + kseg) : zero_bf16x8();
            long c = col0 + i * 16 + frow;
            b[i] = (c < N) ? load_bf16x8(Wt + c * K + k0 + kseg) : zero_bf16x8();
        }
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a[i], b[j], acc[i][j], 0, 0, 0);
    }

    // C/D layout: col = lane&15, row = (lane>>4)*4 + r
    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long rr = row0 + i * 16 + crow + r;
                long cc = col0 + j * 16 + ccol;
                if (rr < M && cc < N) {
                    float v = acc[i][j][r];
                    if (HAS_BIAS) v += bias[cc];
                    if (ACT == 1) v = fmaxf(v, 0.f);
                    if (OUT_F32)
                        reinterpret_cast<float*>(out)[rr * N + cc] = v;
                    else
                        reinterpret_cast<__hip_bfloat16*>(out)[rr * N + cc] =
                            f2bf(v);
                }
            }
}

// ---------------------------------------------------------------------------
// gemm_bias_act2: fat-tile variant for the big GEMMs (FC 3136->512, LSTM
// gate GEMMs 544->2048, head hidden 512->512).  128x128 block, each wave a
// 64x64 tile of 4x4 fragments: 16 MFMAs per 8 operand loads per k-chunk
// (vs 4 per 4 in the 64x64 kernel) — doubles the MFMA issue density while
// operands stream from L2.  Same accumulation order per output element as
// the small kernel (bitwise-identical results).
// ---------------------------------------------------------------------------
template <int ACT, bool HAS_BIAS, bool OUT_F32>
__global__ __launch_bounds__(256) void gemm_bias_act2_kernel(
    const __hip_bfloat16* __restrict__ A,   // (M, K)
    const __hip_bfloat16* __restrict__ Wt,  // (N, K)
    const float* __restrict__ bias,         // (N,)
    void* __restrict__ out,                 // (M, N)
    int M, int N, int K) {
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave >> 1, wc = wave & 1;
    long row0 = (long)blockIdx.x * 128 + wr * 64;
    long col0 = (long)blockIdx.y * 128 + wc * 64;
    int frow = lane & 15;
    int kseg = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        bf16x8 a[4], b[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            long r = row0 + i * 16 + frow;
            a[i] = (r < M) ? load_bf16x8(A + r * K + k0 + kseg) : zero_bf16x8();
            long c = col0 + i * 16 + frow;
            b[i] = (c < N) ? load_bf16x8(Wt + c * K + k0 + kseg) : zero_bf16x8();
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a[i], b[j], acc[i][j], 0, 0, 0);
    }

    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long rr = row0 + i * 16 + crow + r;
                long cc = col0 + j * 16 + ccol;
                if (rr < M && cc < N) {
                    float v = acc[i][j][r];
                    if (HAS_BIAS) v += bias[cc];
                    if (ACT == 1) v = fmaxf(v, 0.f);
                    if (OUT_F32)
                        reinterpret_cast<float*>(out)[rr * N + cc] = v;
                    else
                        reinterpret_cast<__hip_bfloat16*>(out)[rr * N + cc] =
                            f2bf(v);
                }
            }
}

// ---------------------------------------------------------------------------
// gemm_dgrad: dA(M,K) = dY(M,N) @ W(K,N)^T-with-W-stored-(K,N)... i.e.
//   dA[m][k] = sum_n dY[m][n] * W[k][n], W prepacked row-major (K, N).
//   RELU_MASK: multiply dY by (act_out > 0) on load (fused ReLU backward,
//   act_out is the forward output of this layer, same shape as dY).
//   OUT_MASK: multiply dA by (outm > 0) on store — fuses the ReLU backward
//   of the layer BELOW (e.g. conv3's mask applied to the FC dgrad output),
//   so no separate elementwise mask pass runs on the big dX tensors.
// ---------------------------------------------------------------------------
template <bool RELU_MASK, bool OUT_MASK = false>
__global__ __launch_bounds__(256) void gemm_dgrad_kernel(
    const __hip_bfloat16* __restrict__ dY,      // (M, N)
    const __hip_bfloat16* __restrict__ act,     // (M, N) or null
    const __hip_bfloat16* __restrict__ W,       // (K, N) row-major
    __hip_bfloat16* __restrict__ dA,            // (M, K)
    const __hip_bfloat16* __restrict__ outm,    // (M, Kout) or null
    int M, int N, int K, int Kout) {
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave >> 1, wc = wave & 1;
    long row0 = (long)blockIdx.x * 64 + wr * 32;   // m
    long col0 = (long)blockIdx.y * 64 + wc * 32;   // k (output feature dim)
    int frow = lane & 15;
    int nseg = (lane >> 4) * 8;

    f32x4 acc[2][2] = {};
    for (int n0 = 0; n0 < N; n0 += 32) {
        bf16x8 a[2], b[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            long r = row0 + i * 16 + frow;
            if (r < M) {
                a[i] = load_bf16x8(dY + r * N + n0 + nseg);
                if (RELU_MASK) {
                    bf16x8 mv = load_bf16x8(act + r * N + n0 + nseg);
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float vv = (float)a[i][e];
                        a[i][e] = (__bf16)(((float)mv[e] > 0.f) ? vv : 0.f);
                    }
                }
            } else {
                a[i] = zero_bf16x8();
            }
            long c = col0 + i * 16 + frow;
            b[i] = (c < K) ? load_bf16x8(W + c * N + n0 + nseg) : zero_bf16x8();
        }
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a[i], b[j], acc[i][j], 0, 0, 0);
    }
    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long rr = row0 + i * 16 + crow + r;
                long cc = col0 + j * 16 + ccol;
                if (rr < M && cc < Kout) {
                    float v = acc[i][j][r];
                    if (OUT_MASK)
                        v = (bf2f(outm[rr * Kout + cc]) > 0.f) ? v : 0.f;
                    dA[rr * Kout + cc] = f2bf(v);
                }
            }
}

// fat-tile dgrad (same 128x128 / 4x4-fragment scheme as gemm_bias_act2)
template <bool RELU_MASK, bool OUT_MASK = false>
__global__ __launch_bounds__(256) void gemm_dgrad2_kernel(
    const __hip_bfloat16* __restrict__ dY,      // (M, N)
    const __hip_bfloat16* __restrict__ act,     // (M, N) or null
    const __hip_bfloat16* __restrict__ W,       // (K, N) row-major
    __hip_bfloat16* __restrict__ dA,            // (M, Kout)
    const __hip_bfloat16* __restrict__ outm,    // (M, Kout) or null
    int M, int N, int K, int Kout) {
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave >> 1, wc = wave & 1;
    long row0 = (long)blockIdx.x * 128 + wr * 64;
    long col0 = (long)blockIdx.y * 128 + wc * 64;
    int frow = lane & 15;
    int nseg = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};
    for (int n0 = 0; n0 < N; n0 += 32) {
        bf16x8 a[4], b[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            long r = row0 + i * 16 + frow;
            if (r < M) {
                a[i] = load_bf16x8(dY + r * N + n0 + nseg);
                if (RELU_MASK) {
                    bf16x8 mv = load_bf16x8(act + r * N + n0 + nseg);
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float vv = (float)a[i][e];
                        a[i][e] = (__bf16)(((float)mv[e] > 0.f) ? vv : 0.f);
                    }
                }
            } else {
                a[i] = zero_bf16x8();
            }
            long c = col0 + i * 16 + frow;
            b[i] = (c < K) ? load_bf16x8(W + c * N + n0 + nseg) : zero_bf16x8();
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a[i], b[j], acc[i][j], 0, 0, 0);
    }
    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long rr = row0 + i * 16 + crow + r;
                long cc = col0 + j * 16 + ccol;
                if (rr < M && cc < Kout) {
                    float v = acc[i][j][r];
                    if (OUT_MASK)
                        v = (bf2f(outm[rr * Kout + cc]) > 0.f) ? v : 0.f;
                    dA[rr * Kout + cc] = f2bf(v);
                }
            }
}

// ---------------------------------------------------------------------------
// gemm_wgrad: dWt(N,K) += dY(M,N)^T @ A(M,K), reduction over M with
// LDS-staged 32-row tiles; row-chunked grid with f32 atomics into dW.
// Also accumulates db(N) = sum_m dY[m][n] when HAS_BIAS.
//   RELU_MASK as in dgrad (applied to dY).
// ---------------------------------------------------------------------------
// KPERM: the k axis is a packed (d0, d1, d2) index (e.g. conv taps (ky, kx,
// cin) or the FC's flattened (h, w, c)); remap each store to torch's
// (d2, d0, d1) order so gradients accumulate STRAIGHT into the module
// .grad views with no per-step permute-copy kernels.
template <bool RELU_MASK, bool HAS_BIAS, bool KPERM = false>
__global__ __launch_bounds__(256) void gemm_wgrad_kernel(
    const __hip_bfloat16* __restrict__ dY,   // (M, N)
    const __hip_bfloat16* __restrict__ act,  // (M, N) or null
    const __hip_bfloat16* __restrict__ A,    // (M, K)
    float* __restrict__ dWt,                 // (N, K) f32 accumulate
    float* __restrict__ db,                  // (N,) f32 accumulate
    int M, int N, int K, int rows_per_chunk, int Nout, int Kout,
    int kd0 = 0, int kd1 = 0, int kd2 = 0) {
    // grid.x: row chunks; grid.y: N tiles of 64; grid.z: K tiles of 64
    __shared__ __hip_bfloat16 s_dy[32][64 + 8];  // [m][n]
    __shared__ __hip_bfloat16 s_a[32][64 + 8];   // [m][k]
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave >> 1, wc = wave & 1;
    long mstart = (long)blockIdx.x * rows_per_chunk;
    long mend = min((long)M, mstart + rows_per_chunk);
    long ncol0 = (long)blockIdx.y * 64;
    long kcol0 = (long)blockIdx.z * 64;

    int frow = lane & 15;
    int mseg = (lane >> 4) * 8;

    f32x4 acc[2][2] = {};
    float bias_acc = 0.f;   // lane-partial db for column ncol0 + (tid%64)

    for (long m0 = mstart; m0 < mend; m0 += 32) {
        // stage 32 rows x 64 cols of dY and A into LDS (coalesced)
        __syncthreads();
        // 256 threads, each loads 8 elements: 32*64/8 = 256 loads per tile
        {
            int t = threadIdx.x;
            int mrow = t / 8;           // 0..31
            int ncol = (t % 8) * 8;     // 0..56
            long gm = m0 + mrow;
            bf16x8 v = zero_bf16x8();
            if (gm < mend) {
                if (ncol0 + ncol + 8 <= N) {
                    v = load_bf16x8(dY + gm * N + ncol0 + ncol);
                    if (RELU_MASK) {
                        bf16x8 mv = load_bf16x8(act + gm * N + ncol0 + ncol);
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float vv = (float)v[e];
                            v[e] = (__bf16)(((float)mv[e] > 0.f) ? vv : 0.f);
                        }
                    }
                } else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        long c = ncol0 + ncol + e;
                        float vv = (c < N) ? (float)((const __bf16*)dY)[gm * N + c]
                                           : 0.f;
                        if (RELU_MASK) {
                            float m_ = (c < N)
                                ? (float)((const __bf16*)act)[gm * N + c] : 0.f;
                            vv = (m_ > 0.f) ? vv : 0.f;
                        }
                        v[e] = (__bf16)vv;
                    }
                }
            }
            *reinterpret_cast<bf16x8*>(&s_dy[mrow][ncol]) = v;
            bf16x8 w = zero_bf16x8();
            if (gm < mend) {
                if (kcol0 + ncol + 8 <= K)
                    w = load_bf16x8(A + gm * K + kcol0 + ncol);
                else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        long c = kcol0 + ncol + e;
                        if (c < K) w[e] = ((const __bf16*)A)[gm * K + c];
                    }
                }
            }
            *reinterpret_cast<bf16x8*>(&s_a[mrow][ncol]) = w;
        }
        __syncthreads();

        // MFMA: out tile (n, k); gemm-M = n-dim, gemm-N = k-dim, gemm-K = m
        // A-frag: s_dy column (n = row0 + frow), 8 m values
        // B-frag: s_a column (k), 8 m values — both gathered by hardware
        // transpose-reads (ds_read_b64_tr_b16, common.h lds_col_frag8)
        {
            bf16x8 fa[2], fb[2];
#pragma unroll
            for (int i = 0; i < 2; ++i) {
                fa[i] = lds_col_frag8<64 + 8>(&s_dy[0][0], mseg,
                                              wr * 32 + i * 16, lane);
                fb[i] = lds_col_frag8<64 + 8>(&s_a[0][0], mseg,
                                              wc * 32 + i * 16, lane);
            }
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        fa[i], fb[j], acc[i][j], 0, 0, 0);
        }
        if (HAS_BIAS) {
            // db: threads 0..63 own column tid%64; sum the 32 rows staged
            int c = threadIdx.x % 64;
            if (threadIdx.x < 64) {
                for (int mr = 0; mr < 32; ++mr)
                    bias_acc += (float)s_dy[mr][c];
            }
        }
    }

    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long nn = ncol0 + wr * 32 + i * 16 + crow + r;
                long kk = kcol0 + wc * 32 + j * 16 + ccol;
                if (nn < Nout && kk < Kout) {
                    if (KPERM) {
                        int d2 = (int)(kk % kd2);
                        int t_ = (int)(kk / kd2);
                        int d1 = t_ % kd1;
                        int d0 = t_ / kd1;
                        kk = ((long)d2 * kd0 + d0) * kd1 + d1;
                    }
                    atomicAdd(&dWt[nn * Kout + kk], acc[i][j][r]);
                }
            }
    if (HAS_BIAS && threadIdx.x < 64 && blockIdx.z == 0) {
        long c = ncol0 + threadIdx.x;
        if (c < Nout) atomicAdd(&db[c], bias_acc);
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static inline int cdiv(long a, long b) { return (int)((a + b - 1) / b); }

torch::Tensor gemm_bias_act(torch::Tensor A, torch::Tensor Wt,
                            torch::Tensor bias, int64_t act, bool out_f32) {
    TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16 && A.is_contiguous());
    TORCH_CHECK(Wt.dtype() == torch::kBFloat16 && Wt.is_contiguous());
    long M = A.size(0), K = A.size(1), N = Wt.size(0);
    TORCH_CHECK(Wt.size(1) == K && K % 32 == 0, "K must be a multiple of 32");
    bool has_bias = bias.defined() && bias.numel() > 0;
    auto out = torch::empty({M, N}, A.options().dtype(
        out_f32 ? torch::kFloat32 : torch::kBFloat16));
    // fat-tile kernel for shapes big enough to fill the chip with 128x128
    // blocks (the small-N head-output GEMMs keep the 64x64 kernel)
    bool fat = (M >= 1024 && N >= 128);
    dim3 grid(cdiv(M, fat ? 128 : 64), cdiv(N, fat ? 128 : 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    const float* bptr = has_bias ? bias.data_ptr<float>() : nullptr;
    auto* a = reinterpret_cast<const __hip_bfloat16*>(A.data_ptr());
    auto* w = reinterpret_cast<const __hip_bfloat16*>(Wt.data_ptr());

#define LAUNCH(ACT, HB, OF)                                                    \
    do {                                                                       \
        if (fat)                                                               \
            hipLaunchKernelGGL((gemm_bias_act2_kernel<ACT, HB, OF>), grid,     \
                               dim3(256), 0, stream.stream(), a, w, bptr,      \
                               out.data_ptr(), (int)M, (int)N, (int)K);        \
        else                                                                   \
            hipLaunchKernelGGL((gemm_bias_act_kernel<ACT, HB, OF>), grid,      \
                               dim3(256), 0, stream.stream(), a, w, bptr,      \
                               out.data_ptr(), (int)M, (int)N, (int)K);        \
    } while (0)
    if (has_bias) {
        if (act == 1) { if (out_f32) LAUNCH(1, true, true); else LAUNCH(1, true, false); }
        else          { if (out_f32) LAUNCH(0, true, true); else LAUNCH(0, true, false); }
    } else {
        if (act == 1) { if (out_f32) LAUNCH(1, false, true); else LAUNCH(1, false, false); }
        else          { if (out_f32) LAUNCH(0, false, true); else LAUNCH(0, false, false); }
    }
#undef LAUNCH
    return out;
}

torch::Tensor gemm_dgrad(torch::Tensor dY, torch::Tensor act_out,
                         torch::Tensor W, bool relu_mask, int64_t k_out,
                         c10::optional<torch::Tensor> out_mask_opt) {
    torch::Tensor out_mask =
        out_mask_opt.has_value() ? *out_mask_opt : torch::Tensor();
    TORCH_CHECK(dY.is_cuda() && dY.dtype() == torch::kBFloat16 && dY.is_contiguous());
    TORCH_CHECK(W.dtype() == torch::kBFloat16 && W.is_contiguous());
    long M = dY.size(0), N = dY.size(1), K = W.size(0);
    TORCH_CHECK(W.size(1) == N && N % 32 == 0, "N must be a multiple of 32");
    // k_out: emit only the first k_out input-feature columns as a DENSE
    // (M, k_out) tensor (e.g. the latent slice of the padded LSTM input)
    long Kout = (k_out > 0) ? k_out : K;
    TORCH_CHECK(Kout <= K);
    bool has_om = out_mask.defined() && out_mask.numel() > 0;
    if (has_om)
        TORCH_CHECK(out_mask.dtype() == torch::kBFloat16
                    && out_mask.is_contiguous()
                    && out_mask.numel() == M * Kout);
    auto dA = torch::empty({M, Kout}, dY.options());
    bool fat = (M >= 1024 && Kout >= 128);
    dim3 grid(cdiv(M, fat ? 128 : 64), cdiv(Kout, fat ? 128 : 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* ac = relu_mask
        ? reinterpret_cast<const __hip_bfloat16*>(act_out.data_ptr()) : nullptr;
    auto* w = reinterpret_cast<const __hip_bfloat16*>(W.data_ptr());
    auto* da = reinterpret_cast<__hip_bfloat16*>(dA.data_ptr());
    auto* om = has_om
        ? reinterpret_cast<const __hip_bfloat16*>(out_mask.data_ptr()) : nullptr;
#define LAUNCHD(RM, OM)                                                        \
    do {                                                                       \
        if (fat)                                                               \
            hipLaunchKernelGGL((gemm_dgrad2_kernel<RM, OM>), grid, dim3(256),  \
                               0, stream.stream(), dy, ac, w, da, om, (int)M,  \
                               (int)N, (int)K, (int)Kout);                     \
        else                                                                   \
            hipLaunchKernelGGL((gemm_dgrad_kernel<RM, OM>), grid, dim3(256),   \
                               0, stream.stream(), dy, ac, w, da, om, (int)M,  \
                               (int)N, (int)K, (int)Kout);                     \
    } while (0)
    if (relu_mask) { if (has_om) LAUNCHD(true, true); else LAUNCHD(true, false); }
    else           { if (has_om) LAUNCHD(false, true); else LAUNCHD(false, false); }
#undef LAUNCHD
    return dA;
}

std::vector<torch::Tensor> gemm_wgrad(torch::Tensor dY, torch::Tensor act_out,
                                      torch::Tensor A, bool relu_mask,
                                      bool want_bias) {
    TORCH_CHECK(dY.is_cuda() && dY.dtype() == torch::kBFloat16 && dY.is_contiguous());
    TORCH_CHECK(A.dtype() == torch::kBFloat16 && A.is_contiguous());
    long M = dY.size(0), N = dY.size(1), K = A.size(1);
    auto dWt = torch::zeros({N, K}, dY.options().dtype(torch::kFloat32));
    auto db = torch::zeros({want_bias ? N : 1},
                           dY.options().dtype(torch::kFloat32));
    // chunk rows so ~1024 blocks exist for the z=0 plane
    long tiles = (long)cdiv(N, 64) * cdiv(K, 64);
    long target_chunks = std::max(1L, 1024L / std::max(1L, tiles));
    long rows_per_chunk = std::max(32L, (M + target_chunks - 1) / target_chunks);
    rows_per_chunk = ((rows_per_chunk + 31) / 32) * 32;
    dim3 grid(cdiv(M, rows_per_chunk), cdiv(N, 64), cdiv(K, 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* ac = relu_mask
        ? reinterpret_cast<const __hip_bfloat16*>(act_out.data_ptr()) : nullptr;
    auto* a = reinterpret_cast<const __hip_bfloat16*>(A.data_ptr());
#define LAUNCHW(RM, HB)                                                        \
    hipLaunchKernelGGL((gemm_wgrad_kernel<RM, HB>), grid, dim3(256), 0,        \
                       stream.stream(), dy, ac, a, dWt.data_ptr<float>(),      \
                       db.data_ptr<float>(), (int)M, (int)N, (int)K,           \
                       (int)rows_per_chunk, (int)N, (int)K, 0, 0, 0)
    if (relu_mask) { if (want_bias) LAUNCHW(true, true); else LAUNCHW(true, false); }
    else           { if (want_bias) LAUNCHW(false, true); else LAUNCHW(false, false); }
#undef LAUNCHW
    return {dWt, db};
}

// accumulate straight into pre-zeroed .grad tensors (possibly narrower than
// the padded compute width: Nout <= N, Kout <= K) — removes the per-step
// grad-copy kernels from the engine's backward.
void gemm_wgrad_into(torch::Tensor dY, torch::Tensor act_out, torch::Tensor A,
                     bool relu_mask, torch::Tensor dW_out,
                     torch::Tensor db_out,
                     int64_t kd0, int64_t kd1, int64_t kd2) {
    TORCH_CHECK(dY.is_cuda() && dY.dtype() == torch::kBFloat16 && dY.is_contiguous());
    TORCH_CHECK(A.dtype() == torch::kBFloat16 && A.is_contiguous());
    TORCH_CHECK(dW_out.dtype() == torch::kFloat32 && dW_out.is_contiguous());
    long M = dY.size(0), N = dY.size(1), K = A.size(1);
    long Nout = dW_out.size(0), Kout = dW_out.numel() / Nout;
    bool want_bias = db_out.defined() && db_out.numel() > 0;
    bool kperm = kd2 > 0;
    if (kperm) TORCH_CHECK(kd0 * kd1 * kd2 == Kout, "kperm dims mismatch");
    long tiles = (long)cdiv(N, 64) * cdiv(K, 64);
    long target_chunks = std::max(1L, 1024L / std::max(1L, tiles));
    long rows_per_chunk = std::max(32L, (M + target_chunks - 1) / target_chunks);
    rows_per_chunk = ((rows_per_chunk + 31) / 32) * 32;
    dim3 grid(cdiv(M, rows_per_chunk), cdiv(N, 64), cdiv(K, 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* ac = relu_mask
        ? reinterpret_cast<const __hip_bfloat16*>(act_out.data_ptr()) : nullptr;
    auto* a = reinterpret_cast<const __hip_bfloat16*>(A.data_ptr());
    float* dbp = want_bias ? db_out.data_ptr<float>() : nullptr;
#define LAUNCHW(RM, HB, KP)                                                    \
    hipLaunchKernelGGL((gemm_wgrad_kernel<RM, HB, KP>), grid, dim3(256), 0,    \
                       stream.stream(), dy, ac, a, dW_out.data_ptr<float>(),   \
                       dbp, (int)M, (int)N, (int)K,                            \
                       (int)rows_per_chunk, (int)Nout, (int)Kout,              \
                       (int)kd0, (int)kd1, (int)kd2)
#define LAUNCHW2(RM, HB)                                                       \
    do { if (kperm) LAUNCHW(RM, HB, true); else LAUNCHW(RM, HB, false); } while (0)
    if (relu_mask) { if (want_bias) LAUNCHW2(true, true); else LAUNCHW2(true, false); }
    else           { if (want_bias) LAUNCHW2(false, true); else LAUNCHW2(false, false); }
#undef LAUNCHW2
#undef LAUNCHW
}
