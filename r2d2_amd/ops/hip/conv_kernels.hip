// Implicit-GEMM MFMA convolution kernels for the Nature-CNN encoder (gfx950).
//
// Replaces MIOpen's conv paths, which on this workload fall back to naive
// f64-accumulation kernels + 261k per-image im2col launches (see
// profiles/r01_eager_step_kernel_stats.csv: 65% of all GPU time).
//
// Layout: NHWC, bf16 activations (uint8 input for conv1 with fused /255
// dequant).  K-order is (ky, kx, c), so a lane's 8 consecutive k-elements
// are 8 CONTIGUOUS input bytes/bf16 (x-then-c is memory-contiguous in NHWC
// and KW*CIN % 8 == 0 for all three convs) — A fragments load straight from
// global, no im2col materialization.  Weights are prepacked (COUT, K) for
// forward / wgrad and (CIN, TAPS*COUT) per tap-class for dgrad.
//
// Forward:  out(M=N*OH*OW, COUT) = act(patch(M,K) @ Wt^T + bias)
// Dgrad:    dX from a zero-padded dY via uniform tap tables (stride-2 convs
//           split into parity classes so every row in a tile has the same
//           taps; no divergence).
// Wgrad:    dWt(COUT,K) += dY^T @ patch with LDS-staged 32-row tiles and
//           f32 atomics; fused ReLU mask + bias grad.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

union cbf8u {
    bf16x8 v;
    uint4 u;
    __bf16 e[8];
    unsigned short s[8];
};

__device__ __forceinline__ bf16x8 cload_bf16x8(const __hip_bfloat16* p) {
    cbf8u r;
    r.u = *reinterpret_cast<const uint4*>(p);
    return r.v;
}

__device__ __forceinline__ bf16x8 czero() {
    cbf8u r;
    r.u = uint4{0, 0, 0, 0};
    return r.v;
}

// load 8 uint8 bytes and dequantize to bf16/255
__device__ __forceinline__ bf16x8 load_dequant8(const unsigned char* p) {
    uint2 raw = *reinterpret_cast<const uint2*>(p);
    bf16x8 r;
    const float inv = 1.f / 255.f;
#pragma unroll
    for (int i = 0; i < 4; ++i)
        r[i] = (__bf16)(((raw.x >> (8 * i)) & 0xff) * inv);
#pragma unroll
    for (int i = 0; i < 4; ++i)
        r[4 + i] = (__bf16)(((raw.y >> (8 * i)) & 0xff) * inv);
    return r;
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
template <bool IN_U8, int KH, int KW, int CIN, int S, int NCOL, bool RELU>
__global__ __launch_bounds__(256) void conv_fwd_kernel(
    const void* __restrict__ in,            // (N, INH, INW, CIN)
    const __hip_bfloat16* __restrict__ Wt,  // (COUT, K)
    const float* __restrict__ bias,         // (COUT,)
    __hip_bfloat16* __restrict__ out,       // (M, COUT)
    int M, int INH, int INW, int OH, int OW, int COUT) {
    constexpr int K = KH * KW * CIN;
    constexpr int KWC = KW * CIN;
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    // NCOL == 32 (e.g. conv1 COUT=32): 4 waves stack on the M dim so no
    // wave computes guarded-away columns; else classic 2x2
    int wr = (NCOL == 32) ? wave : (wave >> 1);
    int wc = (NCOL == 32) ? 0 : (wave & 1);
    long row0 = (long)blockIdx.x * (NCOL == 32 ? 128 : 64) + wr * 32;
    long col0 = (long)blockIdx.y * 64 + wc * 32;
    int frow = lane & 15;
    int kseg = (lane >> 4) * 8;

    // per-lane patch base addresses for its two A rows
    // 32-bit index math: M <= B*T*84*84 < 2^31, so unsigned division
    // lowers to multiply-shift instead of 64-bit libcalls
    const unsigned OHW = (unsigned)(OH * OW);
    long abase[2];
    bool avalid[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        unsigned r = (unsigned)row0 + i * 16 + frow;
        avalid[i] = r < (unsigned)M;
        if (avalid[i]) {
            unsigned n = r / OHW;
            unsigned p = r % OHW;
            int oy = p / (unsigned)OW, ox = p % (unsigned)OW;
            abase[i] = (((long)n * INH + (long)oy * S) * INW
                        + (long)ox * S) * CIN;
        } else {
            abase[i] = 0;
        }
    }

    f32x4 acc[2][2] = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        int k = k0 + kseg;
        int dy = k / KWC;
        int rem = k % KWC;
        long off = (long)dy * INW * CIN + rem;
        bf16x8 a[2], b[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            if (avalid[i]) {
                if (IN_U8)
                    a[i] = load_dequant8(
                        reinterpret_cast<const unsigned char*>(in) + abase[i] + off);
                else
                    a[i] = cload_bf16x8(
                        reinterpret_cast<const __hip_bfloat16*>(in) + abase[i] + off);
            } else {
                a[i] = czero();
            }
            long c = col0 + i * 16 + frow;
            b[i] = (c < COUT) ? cload_bf16x8(Wt + c * K + k0 + kseg) : czero();
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
                if (rr < M && cc < COUT) {
                    float v = acc[i][j][r] + bias[cc];
                    if (RELU) v = fmaxf(v, 0.f);
                    out[rr * COUT + cc] = f2bf(v);
                }
            }
}

// ---------------------------------------------------------------------------
// Dgrad: dX[n, y, x, ci] = sum_t sum_co dYp[n, (y-dy_t)/S + pad, ...] *
//                          Wd[ci][t*COUT + co]
// Launched per tap-class; rows enumerate the class's (n, yy, xx) grid with
// y = y0 + yy*S.  dYp is the zero-PADDED upstream gradient (pre-masked by
// the ReLU of this conv's output).
// ---------------------------------------------------------------------------
// CO_T: compile-time COUT (both nature dgrads have COUT=64) — the k ->
// (tap, co) split per lane per k-iteration otherwise runs runtime
// divisions (PMC: 72:1 VALU:MFMA on this kernel).
template <int TAPS, int NCOL, int CO_T = 0>
__global__ __launch_bounds__(256) void conv_dgrad_kernel(
    const __hip_bfloat16* __restrict__ dYp,  // (N, PH, PW, COUT)
    const __hip_bfloat16* __restrict__ Wd,   // (CIN, TAPS*COUT)
    __hip_bfloat16* __restrict__ dX,         // (N, XH, XW, CIN)
    const int* __restrict__ taps,            // (TAPS, 2): dy, dx
    int Mc, int YY, int XX, int y0, int x0, int S, int pad,
    int PH, int PW, int COUT, int XH, int XW, int CIN) {
    const int COUTc = CO_T ? CO_T : COUT;
    const int K = TAPS * COUTc;
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = (NCOL == 32) ? wave : (wave >> 1);
    int wc = (NCOL == 32) ? 0 : (wave & 1);
    long row0 = (long)blockIdx.x * (NCOL == 32 ? 128 : 64) + wr * 32;
    long col0 = (long)blockIdx.y * 64 + wc * 32;
    int frow = lane & 15;
    int kseg = (lane >> 4) * 8;

    // per-lane (n, y, x) for its two rows (32-bit division)
    const unsigned YX = (unsigned)(YY * XX);
    long nbase[2];
    int yv[2], xv[2];
    bool avalid[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        unsigned r = (unsigned)row0 + i * 16 + frow;
        avalid[i] = r < (unsigned)Mc;
        unsigned n = avalid[i] ? r / YX : 0;
        unsigned p = avalid[i] ? r % YX : 0;
        yv[i] = y0 + (int)(p / (unsigned)XX) * S;
        xv[i] = x0 + (int)(p % (unsigned)XX) * S;
        nbase[i] = (long)n * PH * PW;
    }

    f32x4 acc[2][2] = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        int k = k0 + kseg;
        int t = (unsigned)k / (unsigned)COUTc;
        int co = (unsigned)k % (unsigned)COUTc;
        int dy = taps[2 * t], dx = taps[2 * t + 1];
        bf16x8 a[2], b[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            if (avalid[i]) {
                int oy = (yv[i] - dy) / S + pad;
                int ox = (xv[i] - dx) / S + pad;
                a[i] = cload_bf16x8(dYp + (nbase[i] + (long)oy * PW + ox) * COUT + co);
            } else {
                a[i] = czero();
            }
            long c = col0 + i * 16 + frow;
            b[i] = (c < CIN) ? cload_bf16x8(Wd + c * K + k0 + kseg) : czero();
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
                if (rr < Mc && cc < CIN) {
                    unsigned n = (unsigned)rr / YX;
                    unsigned p = (unsigned)rr % YX;
                    int y = y0 + (int)(p / (unsigned)XX) * S;
                    int x = x0 + (int)(p % (unsigned)XX) * S;
                    dX[(((long)n * XH + y) * XW + x) * CIN + cc] =
                        f2bf(acc[i][j][r]);
                }
            }
}

// ---------------------------------------------------------------------------
// Dgrad from the DENSE (unpadded, pre-masked) upstream gradient
// dY (N, OH, OW, COUT): taps landing outside [0,OH)x[0,OW) contribute zero
// via a bounds check instead of a zero-padded staging copy (drops the
// dyp zeros-fill + interior copy + separate ReLU-mask pass from the engine
// backward).  OUT_MASK applies the ReLU mask of the conv BELOW
// ((act_x > 0) at the dX address) on store, so the next layer's dgrad and
// wgrad consume a pre-masked tensor directly.
// ---------------------------------------------------------------------------
// GT (geometry template): nonzero = compile-time (OH==OW, XH==XW, S) for
// the hot Nature shapes — the per-lane row->(n,y,x) and tap divisions
// lower to mul-shift by constants (this kernel is otherwise VALU-bound on
// address math, r06 PMC 72:1 VALU:MFMA).  Encoded GT = OH*1000 + XH*10 + S.
template <int TAPS, int NCOL, int CO_T, bool OUT_MASK, int GT = 0>
__global__ __launch_bounds__(256) void conv_dgrad_dense_kernel(
    const __hip_bfloat16* __restrict__ dY,   // (N, OH, OW, COUT)
    const __hip_bfloat16* __restrict__ Wd,   // (CIN, TAPS*COUT)
    const __hip_bfloat16* __restrict__ actx, // (N, XH, XW, CIN) or null
    __hip_bfloat16* __restrict__ dX,         // (N, XH, XW, CIN)
    const int* __restrict__ taps,            // (TAPS, 2): dy, dx
    int Mc, int YY_, int XX_, int y0, int x0, int S_,
    int OH_, int OW_, int COUT, int XH_, int XW_, int CIN) {
    const int S = GT ? (GT % 10) : S_;
    const int OH = GT ? (GT / 1000) : OH_;
    const int OW = GT ? (GT / 1000) : OW_;
    const int XH = GT ? ((GT / 10) % 100) : XH_;
    const int XW = GT ? ((GT / 10) % 100) : XW_;
    // for both hot geometries XH/S equals the class grid extent for every
    // parity offset (20/2=10, 9/1=9), keeping YY/XX compile-time under GT
    const int YY = GT ? (XH / S) : YY_;
    const int XX = GT ? (XW / S) : XX_;
    const int COUTc = CO_T ? CO_T : COUT;
    const int K = TAPS * COUTc;
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = (NCOL == 32) ? wave : (wave >> 1);
    int wc = (NCOL == 32) ? 0 : (wave & 1);
    long row0 = (long)blockIdx.x * (NCOL == 32 ? 128 : 64) + wr * 32;
    long col0 = (long)blockIdx.y * 64 + wc * 32;
    int frow = lane & 15;
    int kseg = (lane >> 4) * 8;

    const unsigned YX = (unsigned)(YY * XX);
    long nbase[2];
    int yv[2], xv[2];
    bool avalid[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        unsigned r = (unsigned)row0 + i * 16 + frow;
        avalid[i] = r < (unsigned)Mc;
        unsigned n = avalid[i] ? r / YX : 0;
        unsigned p = avalid[i] ? r % YX : 0;
        yv[i] = y0 + (int)(p / (unsigned)XX) * S;
        xv[i] = x0 + (int)(p % (unsigned)XX) * S;
        nbase[i] = (long)n * OH * OW;
    }

    f32x4 acc[2][2] = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        int k = k0 + kseg;
        int t = (unsigned)k / (unsigned)COUTc;
        int co = (unsigned)k % (unsigned)COUTc;
        int dy = taps[2 * t], dx = taps[2 * t + 1];
        bf16x8 a[2], b[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            // parity class guarantees S | (yv-dy); bounds replace padding
            int ny = yv[i] - dy, nx = xv[i] - dx;
            int oy = ny / S, ox = nx / S;
            bool ok = avalid[i] && ny >= 0 && nx >= 0 && oy < OH && ox < OW;
            a[i] = ok
                ? cload_bf16x8(dY + (nbase[i] + (long)oy * OW + ox) * COUT + co)
                : czero();
            long c = col0 + i * 16 + frow;
            b[i] = (c < CIN) ? cload_bf16x8(Wd + c * K + k0 + kseg) : czero();
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
                if (rr < Mc && cc < CIN) {
                    unsigned n = (unsigned)rr / YX;
                    unsigned p = (unsigned)rr % YX;
                    int y = y0 + (int)(p / (unsigned)XX) * S;
                    int x = x0 + (int)(p % (unsigned)XX) * S;
                    long addr = (((long)n * XH + y) * XW + x) * CIN + cc;
                    float v = acc[i][j][r];
                    if (OUT_MASK)
                        v = (bf2f(actx[addr]) > 0.f) ? v : 0.f;
                    dX[addr] = f2bf(v);
                }
            }
}

// ---------------------------------------------------------------------------
// Wgrad: dWt(COUT, K) += sum_rows relu_mask(dY)[row][co] * patch[row][k]
// The FULL K extent (<= 576) and full COUT (<= 64) are staged per 32-row
// tile, so dY and the patches are each read exactly once; per-wave
// accumulators cover all k-tiles and are flushed with f32 atomics once per
// chunk.  Fused ReLU mask + bias grad.
// ---------------------------------------------------------------------------
// TORCH_LAYOUT: remap each (ky, kx, ci)-packed k store to torch's
// (ci, ky, kx) weight order, so gradients accumulate straight into the
// module's .grad view (compile-time dims -> cheap index math).
template <bool IN_U8, int KH, int KW, int CIN, int S, int NCOT, bool RELU,
          bool TORCH_LAYOUT = false>
__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const __hip_bfloat16* __restrict__ dY,   // (M, COUT)
    const __hip_bfloat16* __restrict__ act,  // (M, COUT) forward output
    const void* __restrict__ in,             // (N, INH, INW, CIN)
    float* __restrict__ dWt,                 // (COUT, K) f32
    float* __restrict__ db,                  // (COUT,) f32
    int M, int INH, int INW, int OH, int OW, int COUT, int rows_per_chunk) {
    constexpr int K = KH * KW * CIN;
    constexpr int KWC = KW * CIN;
    // NCOT = co tiles of 32 (1 when COUT<=32): waves split K 4/NCOT ways so
    // no wave computes guarded-away co columns
    constexpr int NKW = 4 / NCOT;                     // k splits
    constexpr int KHALF = ((K / NKW + 31) / 32) * 32; // per-wave k extent
    constexpr int KFRAG = KHALF / 16;                 // 16-col frags per wave
    __shared__ __hip_bfloat16 s_dy[32][64 + 8];
    __shared__ __hip_bfloat16 s_a[32][K + 8];
    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = (NCOT == 1) ? 0 : (wave >> 1);           // co tile
    int wc = (NCOT == 1) ? wave : (wave & 1);         // k split
    long mstart = (long)blockIdx.x * rows_per_chunk;
    long mend = min((long)M, mstart + rows_per_chunk);
    int frow = lane & 15;
    int mseg = (lane >> 4) * 8;

    // co extent per wave is always 32 (2 fragments of 16)
    f32x4 acc[2][KFRAG] = {};
    float bias_acc = 0.f;

    for (long m0 = mstart; m0 < mend; m0 += 32) {
        __syncthreads();
        {
            int t = threadIdx.x;
            {
                int mrow = t / 8;
                int col = (t % 8) * 8;
                long gm = m0 + mrow;
                bf16x8 v = czero();
                if (gm < mend) {
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        long c = col + e;
                        float g = (c < COUT) ? bf2f(dY[gm * COUT + c]) : 0.f;
                        if (RELU) {
                            float m_ = (c < COUT) ? bf2f(act[gm * COUT + c]) : 0.f;
                            g = (m_ > 0.f) ? g : 0.f;
                        }
                        v[e] = (__bf16)g;
                    }
                }
                *reinterpret_cast<bf16x8*>(&s_dy[mrow][col]) = v;
            }
            for (int e8 = t; e8 < 32 * (K / 8); e8 += 256) {
                int mrow = e8 / (K / 8);
                int k = (e8 % (K / 8)) * 8;
                long gm = m0 + mrow;
                bf16x8 w = czero();
                if (gm < mend) {
                    unsigned n = (unsigned)gm / (unsigned)(OH * OW);
                    unsigned p = (unsigned)gm % (unsigned)(OH * OW);
                    int oy = p / (unsigned)OW, ox = p % (unsigned)OW;
                    long base = (((long)n * INH + (long)oy * S) * INW
                                 + (long)ox * S) * CIN;
                    int dy_ = k / KWC, rem = k % KWC;
                    long off = base + (long)dy_ * INW * CIN + rem;
                    if (IN_U8)
                        w = load_dequant8(
                            reinterpret_cast<const unsigned char*>(in) + off);
                    else
                        w = cload_bf16x8(
                            reinterpret_cast<const __hip_bfloat16*>(in) + off);
                }
                *reinterpret_cast<bf16x8*>(&s_a[mrow][k]) = w;
            }
        }
        __syncthreads();

        // fragment gathers via hardware transpose-reads (common.h)
        bf16x8 fa[2];
#pragma unroll
        for (int i = 0; i < 2; ++i)
            fa[i] = lds_col_frag8<64 + 8>(&s_dy[0][0], mseg,
                                          wr * 32 + i * 16, lane);
#pragma unroll
        for (int kf = 0; kf < KFRAG; ++kf) {
            int kcol0 = wc * KHALF + kf * 16;
            bf16x8 fb;
            if (kcol0 + 16 <= K) {
                fb = lds_col_frag8<K + 8>(&s_a[0][0], mseg, kcol0, lane);
            } else {                       // partial tail column span
                fb = czero();
                int kcol = kcol0 + frow;
                if (kcol < K) {
#pragma unroll
                    for (int e = 0; e < 8; ++e)
                        fb[e] = *(const __bf16*)&s_a[mseg + e][kcol];
                }
            }
#pragma unroll
            for (int i = 0; i < 2; ++i)
                acc[i][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    fa[i], fb, acc[i][kf], 0, 0, 0);
        }

        if (threadIdx.x < 64) {
            int c = threadIdx.x;
            for (int mr = 0; mr < 32; ++mr) bias_acc += bf2f(s_dy[mr][c]);
        }
    }

    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int kf = 0; kf < KFRAG; ++kf)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long co = wr * 32 + i * 16 + crow + r;
                long kk = wc * KHALF + kf * 16 + ccol;
                if (co < COUT && kk < K) {
                    if (TORCH_LAYOUT) {
                        int ci = (int)(kk % CIN);
                        int t_ = (int)(kk / CIN);
                        int kx = t_ % KW, ky = t_ / KW;
                        kk = ((long)ci * KH + ky) * KW + kx;
                    }
                    atomicAdd(&dWt[co * K + kk], acc[i][kf][r]);
                }
            }
    if (threadIdx.x < 64) {
        long c = threadIdx.x;
        if (c < COUT) atomicAdd(&db[c], bias_acc);
    }
}

// ---------------------------------------------------------------------------
// conv_fwd_band: per-image forward.  The whole input image AND the full
// prepacked weight matrix fit in LDS for every Nature-CNN conv, so the
// patch field reads LDS (one global read + one dequant per input element
// instead of one per patch overlap: conv1 557 MB u8 re-dequant -> 154 MB,
// conv3 9x re-read -> 1x).  Fused bias + ReLU as in conv_fwd_kernel.
// ---------------------------------------------------------------------------
// COUT_T: output columns computed per launch (always 32 — a 64-out conv
// runs as two launches with co0 = 0/32, halving the LDS weight slab so
// 2-3 workgroups fit per CU); COUT_FULL is the output row stride.
template <bool IN_U8, int KH, int KW, int CIN, int S, int INH, int INW,
          int OH, int OW, int COUT_T, int COUT_FULL>
__global__ __launch_bounds__(256) void conv_fwd_band_kernel(
    const void* __restrict__ in,            // (N, INH, INW, CIN)
    const __hip_bfloat16* __restrict__ Wt,  // (COUT_FULL, K)
    const float* __restrict__ bias,
    __hip_bfloat16* __restrict__ out,       // (N*OH*OW, COUT_FULL)
    int N, int imgs_per_wg, int co0) {
    constexpr int K = KH * KW * CIN;
    constexpr int KWC = KW * CIN;
    constexpr int NPIX = OH * OW;
    constexpr int NB = 2;                          // B frags = 32 cols/wave
    constexpr int RPI = 128;                       // 4 waves stack rows

    __shared__ __hip_bfloat16 s_img[INH * INW * CIN];
    __shared__ __hip_bfloat16 s_w[COUT_T][K + 8];

    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = wave;
    int wc = 0;
    int frow = lane & 15;
    int kseg = (lane >> 4) * 8;

    for (int e = threadIdx.x * 8; e < COUT_T * K; e += blockDim.x * 8) {
        int c = e / K, k = e % K;
        *reinterpret_cast<bf16x8*>(&s_w[c][k]) =
            cload_bf16x8(Wt + (long)(co0 + c) * K + k);
    }

    const long n0 = (long)blockIdx.x * imgs_per_wg;
    const long n1 = min((long)N, n0 + imgs_per_wg);
    for (long n = n0; n < n1; ++n) {
        __syncthreads();
        {
            const long gbase = n * INH * INW * CIN;
            for (int e = threadIdx.x * 8; e < INH * INW * CIN;
                 e += blockDim.x * 8) {
                bf16x8 v;
                if (IN_U8)
                    v = load_dequant8(
                        reinterpret_cast<const unsigned char*>(in) + gbase + e);
                else
                    v = cload_bf16x8(
                        reinterpret_cast<const __hip_bfloat16*>(in) + gbase + e);
                *reinterpret_cast<bf16x8*>(&s_img[e]) = v;
            }
        }
        __syncthreads();

        for (int p0 = wr * 32; p0 < NPIX; p0 += RPI) {
            int pbase[2];
            bool pval[2];
#pragma unroll
            for (int i = 0; i < 2; ++i) {
                int pp = p0 + i * 16 + frow;
                pval[i] = pp < NPIX;
                int oy = pp / OW, ox = pp % OW;
                pbase[i] = ((oy * S) * INW + ox * S) * CIN;
            }
            f32x4 acc[2][NB] = {};
            for (int k0 = 0; k0 < K; k0 += 32) {
                int k = k0 + kseg;
                int dy = k / KWC;
                int rem = k % KWC;
                int off = dy * INW * CIN + rem;
                bf16x8 a[2], b[NB];
#pragma unroll
                for (int i = 0; i < 2; ++i)
                    a[i] = pval[i]
                        ? *reinterpret_cast<const bf16x8*>(
                              &s_img[pbase[i] + off])
                        : czero();
#pragma unroll
                for (int j = 0; j < NB; ++j)
                    b[j] = cload_bf16x8(
                        &s_w[wc * 32 + j * 16 + frow][k0 + kseg]);
#pragma unroll
                for (int i = 0; i < 2; ++i)
#pragma unroll
                    for (int j = 0; j < NB; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[i], b[j], acc[i][j], 0, 0, 0);
            }
            int ccol = lane & 15;
            int crow = (lane >> 4) * 4;
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int j = 0; j < NB; ++j)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int pp = p0 + i * 16 + crow + r;
                        int cc = wc * 32 + j * 16 + ccol;
                        if (pp < NPIX) {
                            float v = acc[i][j][r] + bias[co0 + cc];
                            out[(n * NPIX + pp) * COUT_FULL + co0 + cc] =
                                f2bf(fmaxf(v, 0.f));
                        }
                    }
        }
    }
}

// ---------------------------------------------------------------------------
// conv_wgrad_band: per-image wgrad.  The WHOLE input image fits in LDS for
// every Nature-CNN conv (conv1 84x84x4 u8->bf16 56 KB, conv2 20x20x32
// 26 KB, conv3 9x9x64 10 KB), so one workgroup stages it ONCE (single
// dequant per input element instead of one per patch overlap), rebuilds
// each 32-row patch tile LDS->LDS, and accumulates the whole image's
// contribution in registers with ONE atomic flush at the end.  Fragments
// gathered by hardware transpose-reads.  Replaces conv_wgrad_kernel's
// global patch re-reads (conv1: 557 MB -> 154 MB per step).
// ---------------------------------------------------------------------------
template <bool IN_U8, int KH, int KW, int CIN, int S, int INH, int INW,
          int OH, int OW, int COUT_T>
__global__ __launch_bounds__(256) void conv_wgrad_band_kernel(
    const __hip_bfloat16* __restrict__ dY,   // (M=N*OH*OW, COUT) PRE-MASKED
    const __hip_bfloat16* __restrict__ act,  // (M, COUT) forward out (mask)
    const void* __restrict__ in,             // (N, INH, INW, CIN)
    float* __restrict__ dWt,                 // (COUT, K) f32
    float* __restrict__ db,                  // (COUT,) f32
    int N, int imgs_per_wg) {
    constexpr int K = KH * KW * CIN;
    constexpr int KWC = KW * CIN;
    constexpr int NPIX = OH * OW;
    constexpr int NCOT = (COUT_T + 31) / 32;          // cout tiles of 32
    constexpr int NKW = 4 / NCOT;                     // k splits
    constexpr int KHALF = ((K / NKW + 31) / 32) * 32;
    constexpr int KFRAG = KHALF / 16;

    __shared__ __hip_bfloat16 s_img[INH * INW * CIN];
    __shared__ __hip_bfloat16 s_dy[32][64 + 8];
    __shared__ __hip_bfloat16 s_a[32][K + 8];

    int wave = threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    int wr = (NCOT == 1) ? 0 : (wave >> 1);
    int wc = (NCOT == 1) ? wave : (wave & 1);
    int frow = lane & 15;
    int mseg = (lane >> 4) * 8;

    f32x4 acc[2][KFRAG] = {};
    float bias_acc = 0.f;

    const long n0 = (long)blockIdx.x * imgs_per_wg;
    const long n1 = min((long)N, n0 + imgs_per_wg);
    for (long n = n0; n < n1; ++n) {
        __syncthreads();
        {   // stage the whole input image (dequant once per element)
            const long gbase = n * INH * INW * CIN;
            for (int e = threadIdx.x * 8; e < INH * INW * CIN;
                 e += blockDim.x * 8) {
                bf16x8 v;
                if (IN_U8)
                    v = load_dequant8(
                        reinterpret_cast<const unsigned char*>(in) + gbase + e);
                else
                    v = cload_bf16x8(
                        reinterpret_cast<const __hip_bfloat16*>(in) + gbase + e);
                *reinterpret_cast<bf16x8*>(&s_img[e]) = v;
            }
        }
        for (int p0 = 0; p0 < NPIX; p0 += 32) {
            __syncthreads();
            {   // dY rows (pre-masked by the producing kernel; `act` kept
                // for the generic path) + patch rows from the LDS image
                int t = threadIdx.x;
                {
                    int mrow = t / 8;
                    int col = (t % 8) * 8;
                    int pp = p0 + mrow;
                    bf16x8 v = czero();
                    if (pp < NPIX) {
                        long gm = n * NPIX + pp;
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float g = (col + e < COUT_T)
                                ? bf2f(dY[gm * COUT_T + col + e]) : 0.f;
                            float m_ = (col + e < COUT_T)
                                ? bf2f(act[gm * COUT_T + col + e]) : 0.f;
                            v[e] = (__bf16)((m_ > 0.f) ? g : 0.f);
                        }
                    }
                    *reinterpret_cast<bf16x8*>(&s_dy[mrow][col]) = v;
                }
                for (int e8 = t; e8 < 32 * (K / 8); e8 += 256) {
                    int mrow = e8 / (K / 8);
                    int k = (e8 % (K / 8)) * 8;
                    int pp = p0 + mrow;
                    bf16x8 w = czero();
                    if (pp < NPIX) {
                        int oy = pp / OW, ox = pp % OW;
                        int dy_ = k / KWC, rem = k % KWC;
                        w = *reinterpret_cast<const bf16x8*>(
                            &s_img[((oy * S + dy_) * INW + ox * S) * CIN + rem]);
                    }
                    *reinterpret_cast<bf16x8*>(&s_a[mrow][k]) = w;
                }
            }
            __syncthreads();

            bf16x8 fa[2];
#pragma unroll
            for (int i = 0; i < 2; ++i)
                fa[i] = lds_col_frag8<64 + 8>(&s_dy[0][0], mseg,
                                              wr * 32 + i * 16, lane);
#pragma unroll
            for (int kf = 0; kf < KFRAG; ++kf) {
                int kcol0 = wc * KHALF + kf * 16;
                bf16x8 fb;
                if (kcol0 + 16 <= K) {
                    fb = lds_col_frag8<K + 8>(&s_a[0][0], mseg, kcol0, lane);
                } else {
                    fb = czero();
                    int kcol = kcol0 + frow;
                    if (kcol < K) {
#pragma unroll
                        for (int e = 0; e < 8; ++e)
                            fb[e] = *(const __bf16*)&s_a[mseg + e][kcol];
                    }
                }
#pragma unroll
                for (int i = 0; i < 2; ++i)
                    acc[i][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        fa[i], fb, acc[i][kf], 0, 0, 0);
            }
            if (threadIdx.x < 64) {
                int c = threadIdx.x;
                for (int mr = 0; mr < 32; ++mr)
                    bias_acc += bf2f(s_dy[mr][c]);
            }
        }
    }

    int ccol = lane & 15;
    int crow = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int kf = 0; kf < KFRAG; ++kf)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                long co = wr * 32 + i * 16 + crow + r;
                long kk = wc * KHALF + kf * 16 + ccol;
                if (co < COUT_T && kk < K)
                    atomicAdd(&dWt[co * K + kk], acc[i][kf][r]);
            }
    if (threadIdx.x < 64) {
        long c = threadIdx.x;
        if (c < COUT_T) atomicAdd(&db[c], bias_acc);
    }
}

// ---------------------------------------------------------------------------
// Host wrappers.  Conv geometry is dispatched over the three Nature-CNN
// layers (and reusable for any conv matching a template instance).
// ---------------------------------------------------------------------------

static inline int ccdiv(long a, long b) { return (int)((a + b - 1) / b); }

// conv_id: 1 = 8x8 s4 CIN variable(4) u8-in, 2 = 4x4 s2 CIN 32, 3 = 3x3 s1 CIN 64
torch::Tensor conv_fwd(torch::Tensor in, torch::Tensor Wt, torch::Tensor bias,
                       int64_t conv_id, int64_t N, int64_t INH, int64_t INW,
                       int64_t OH, int64_t OW, bool relu) {
    long M = N * OH * OW;
    long COUT = Wt.size(0);
    auto out = torch::empty({M, COUT},
                            in.options().dtype(torch::kBFloat16));
    dim3 grid(ccdiv(M, COUT <= 32 ? 128 : 64), ccdiv(COUT, 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* w = reinterpret_cast<const __hip_bfloat16*>(Wt.data_ptr());
    auto* o = reinterpret_cast<__hip_bfloat16*>(out.data_ptr());
    const float* b = bias.data_ptr<float>();
    const void* x = in.data_ptr();
#define CLAUNCH(U8, KH_, KW_, CIN_, S_)                                        \
    hipLaunchKernelGGL((conv_fwd_kernel<U8, KH_, KW_, CIN_, S_, 64, true>),    \
                       grid,                                                   \
                       dim3(256), 0, stream.stream(), x, w, b, o, (int)M,      \
                       (int)INH, (int)INW, (int)OH, (int)OW, (int)COUT)
    if (conv_id == 1) {
        TORCH_CHECK(in.dtype() == torch::kUInt8);
        hipLaunchKernelGGL((conv_fwd_kernel<true, 8, 8, 4, 4, 32, true>), grid,
                           dim3(256), 0, stream.stream(), x, w, b, o, (int)M,
                           (int)INH, (int)INW, (int)OH, (int)OW, (int)COUT);
    } else if (conv_id == 2) {
        TORCH_CHECK(in.dtype() == torch::kBFloat16);
        CLAUNCH(false, 4, 4, 32, 2);
    } else if (conv_id == 3) {
        CLAUNCH(false, 3, 3, 64, 1);
    } else {
        TORCH_CHECK(false, "unknown conv_id");
    }
#undef CLAUNCH
    return out;
}

torch::Tensor conv_dgrad(torch::Tensor dYp, torch::Tensor Wd, torch::Tensor taps,
                         int64_t N, int64_t PH, int64_t PW, int64_t COUT,
                         int64_t XH, int64_t XW, int64_t CIN,
                         int64_t y0, int64_t x0, int64_t S, int64_t pad,
                         torch::Tensor dX) {
    long YY = (XH - 1 - y0) / S + 1;
    long XX = (XW - 1 - x0) / S + 1;
    long Mc = N * YY * XX;
    long TAPS = taps.size(0);
    dim3 grid(ccdiv(Mc, CIN <= 32 ? 128 : 64), ccdiv(CIN, 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dYp.data_ptr());
    auto* w = reinterpret_cast<const __hip_bfloat16*>(Wd.data_ptr());
    auto* dx = reinterpret_cast<__hip_bfloat16*>(dX.data_ptr());
    const int* tp = taps.data_ptr<int>();
#define DL1(T, NC, CO)                                                         \
    hipLaunchKernelGGL((conv_dgrad_kernel<T, NC, CO>), grid, dim3(256), 0,     \
                       stream.stream(), dy, w, dx, tp, (int)Mc, (int)YY,       \
                       (int)XX, (int)y0, (int)x0, (int)S, (int)pad, (int)PH,   \
                       (int)PW, (int)COUT, (int)XH, (int)XW, (int)CIN)
#define DLAUNCH(T)                                                             \
    if (CIN <= 32) { if (COUT == 64) DL1(T, 32, 64); else DL1(T, 32, 0); }     \
    else { if (COUT == 64) DL1(T, 64, 64); else DL1(T, 64, 0); }
    if (TAPS == 4) { DLAUNCH(4); }
    else if (TAPS == 9) { DLAUNCH(9); }
    else TORCH_CHECK(false, "unsupported tap count");
#undef DLAUNCH
#undef DL1
    return dX;
}

// dense-input dgrad: dY is the UNPADDED, PRE-MASKED upstream gradient
// (N, OH, OW, COUT); actx (optional) fuses the ReLU mask of the conv below
// into the dX store.
torch::Tensor conv_dgrad_dense(torch::Tensor dY, torch::Tensor Wd,
                               torch::Tensor taps, torch::Tensor actx,
                               int64_t N, int64_t OH, int64_t OW, int64_t COUT,
                               int64_t XH, int64_t XW, int64_t CIN,
                               int64_t y0, int64_t x0, int64_t S,
                               torch::Tensor dX) {
    long YY = (XH - 1 - y0) / S + 1;
    long XX = (XW - 1 - x0) / S + 1;
    long Mc = N * YY * XX;
    long TAPS = taps.size(0);
    bool has_m = actx.defined() && actx.numel() > 0;
    if (has_m) TORCH_CHECK(actx.numel() == dX.numel());
    dim3 grid(ccdiv(Mc, CIN <= 32 ? 128 : 64), ccdiv(CIN, 64));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* w = reinterpret_cast<const __hip_bfloat16*>(Wd.data_ptr());
    auto* am = has_m
        ? reinterpret_cast<const __hip_bfloat16*>(actx.data_ptr()) : nullptr;
    auto* dx = reinterpret_cast<__hip_bfloat16*>(dX.data_ptr());
    const int* tp = taps.data_ptr<int>();
#define DD1(T, NC, CO, OM, GT)                                                 \
    hipLaunchKernelGGL((conv_dgrad_dense_kernel<T, NC, CO, OM, GT>), grid,     \
                       dim3(256), 0, stream.stream(), dy, w, am, dx, tp,       \
                       (int)Mc, (int)YY, (int)XX, (int)y0, (int)x0, (int)S,    \
                       (int)OH, (int)OW, (int)COUT, (int)XH, (int)XW, (int)CIN)
#define DDM(T, NC, CO, GT)                                                     \
    do { if (has_m) DD1(T, NC, CO, true, GT);                                  \
         else DD1(T, NC, CO, false, GT); } while (0)
    // compile-time geometry for the hot Nature shapes (address math
    // lowers to mul-shift; GT = OH*1000 + XH*10 + S)
    if (TAPS == 9 && CIN == 64 && COUT == 64 && OH == 7 && XH == 9 && S == 1) {
        DDM(9, 64, 64, 7091);
        return dX;
    }
    if (TAPS == 4 && CIN == 32 && COUT == 64 && OH == 9 && XH == 20 && S == 2) {
        DDM(4, 32, 64, 9202);
        return dX;
    }
#define DDLAUNCH(T)                                                            \
    if (CIN <= 32) { if (COUT == 64) DDM(T, 32, 64, 0); else DDM(T, 32, 0, 0); } \
    else { if (COUT == 64) DDM(T, 64, 64, 0); else DDM(T, 64, 0, 0); }
    if (TAPS == 4) { DDLAUNCH(4); }
    else if (TAPS == 9) { DDLAUNCH(9); }
    else TORCH_CHECK(false, "unsupported tap count");
#undef DDLAUNCH
#undef DDM
#undef DD1
    return dX;
}

// per-image band forward for the three Nature-CNN geometries.
torch::Tensor conv_fwd_band(torch::Tensor in, torch::Tensor Wt,
                            torch::Tensor bias, int64_t conv_id, int64_t N) {
    long COUT = (conv_id == 1) ? 32 : 64;
    long NPIX = (conv_id == 1) ? 400 : (conv_id == 2) ? 81 : 49;
    auto out = torch::empty({N * NPIX, COUT},
                            in.options().dtype(torch::kBFloat16));
    int imgs = (int)((N + 2047) / 2048);
    int grid = (int)((N + imgs - 1) / imgs);
    auto stream = at::cuda::getCurrentCUDAStream();
    const void* x = in.data_ptr();
    auto* w = reinterpret_cast<const __hip_bfloat16*>(Wt.data_ptr());
    const float* b = bias.data_ptr<float>();
    auto* o = reinterpret_cast<__hip_bfloat16*>(out.data_ptr());
#define FBLAUNCH(U8, KH_, KW_, CIN_, S_, INH_, OH_, COF_, CO0_)                \
    hipLaunchKernelGGL((conv_fwd_band_kernel<U8, KH_, KW_, CIN_, S_, INH_,     \
                                             INH_, OH_, OH_, 32, COF_>),       \
                       dim3(grid), dim3(256), 0, stream.stream(), x, w, b, o,  \
                       (int)N, imgs, CO0_)
    if (conv_id == 1) {
        FBLAUNCH(true, 8, 8, 4, 4, 84, 20, 32, 0);
    } else if (conv_id == 2) {   // 64-out: two 32-col launches (LDS halved)
        FBLAUNCH(false, 4, 4, 32, 2, 20, 9, 64, 0);
        FBLAUNCH(false, 4, 4, 32, 2, 20, 9, 64, 32);
    } else if (conv_id == 3) {
        FBLAUNCH(false, 3, 3, 64, 1, 9, 7, 64, 0);
        FBLAUNCH(false, 3, 3, 64, 1, 9, 7, 64, 32);
    } else {
        TORCH_CHECK(false, "unknown conv_id");
    }
#undef FBLAUNCH
    return out;
}

// per-image band wgrad for the three Nature-CNN geometries (compile-time
// shapes; see conv_wgrad_band_kernel).  N = number of images (B*T).
std::vector<torch::Tensor> conv_wgrad_band(torch::Tensor dY, torch::Tensor act,
                                           torch::Tensor in, int64_t conv_id,
                                           int64_t N) {
    long COUT = (conv_id == 1) ? 32 : 64;
    long K = (conv_id == 1) ? 8 * 8 * 4 : (conv_id == 2) ? 4 * 4 * 32
                                                         : 3 * 3 * 64;
    auto dWt = torch::zeros({COUT, K}, dY.options().dtype(torch::kFloat32));
    auto db = torch::zeros({COUT}, dY.options().dtype(torch::kFloat32));
    int imgs = (int)((N + 1023) / 1024);
    int grid = (int)((N + imgs - 1) / imgs);
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* ac = reinterpret_cast<const __hip_bfloat16*>(act.data_ptr());
    const void* x = in.data_ptr();
#define WBLAUNCH(U8, KH_, KW_, CIN_, S_, INH_, OH_, CO_)                       \
    hipLaunchKernelGGL((conv_wgrad_band_kernel<U8, KH_, KW_, CIN_, S_, INH_,   \
                                               INH_, OH_, OH_, CO_>),          \
                       dim3(grid), dim3(256), 0, stream.stream(), dy, ac, x,   \
                       dWt.data_ptr<float>(), db.data_ptr<float>(), (int)N,    \
                       imgs)
    if (conv_id == 1) WBLAUNCH(true, 8, 8, 4, 4, 84, 20, 32);
    else if (conv_id == 2) WBLAUNCH(false, 4, 4, 32, 2, 20, 9, 64);
    else if (conv_id == 3) WBLAUNCH(false, 3, 3, 64, 1, 9, 7, 64);
    else TORCH_CHECK(false, "unknown conv_id");
#undef WBLAUNCH
    return {dWt, db};
}

std::vector<torch::Tensor> conv_wgrad(torch::Tensor dY, torch::Tensor act,
                                      torch::Tensor in, int64_t conv_id,
                                      int64_t N, int64_t INH, int64_t INW,
                                      int64_t OH, int64_t OW, int64_t COUT,
                                      int64_t K) {
    long M = N * OH * OW;
    auto dWt = torch::zeros({COUT, K}, dY.options().dtype(torch::kFloat32));
    auto db = torch::zeros({COUT}, dY.options().dtype(torch::kFloat32));
    long target_chunks = 1024;
    long rows_per_chunk = std::max(32L, (M + target_chunks - 1) / target_chunks);
    rows_per_chunk = ((rows_per_chunk + 31) / 32) * 32;
    dim3 grid(ccdiv(M, rows_per_chunk));
    auto stream = at::cuda::getCurrentCUDAStream();
    auto* dy = reinterpret_cast<const __hip_bfloat16*>(dY.data_ptr());
    auto* ac = reinterpret_cast<const __hip_bfloat16*>(act.data_ptr());
    const void* x = in.data_ptr();
#define WLAUNCH(U8, KH_, KW_, CIN_, S_, NCOT_)                                 \
    hipLaunchKernelGGL((conv_wgrad_kernel<U8, KH_, KW_, CIN_, S_, NCOT_,       \
                                          true>),                              \
                       grid, dim3(256), 0, stream.stream(), dy, ac, x,         \
                       dWt.data_ptr<float>(), db.data_ptr<float>(), (int)M,    \
                       (int)INH, (int)INW, (int)OH, (int)OW, (int)COUT,        \
                       (int)rows_per_chunk)
    if (conv_id == 1) WLAUNCH(true, 8, 8, 4, 4, 1);
    else if (conv_id == 2) WLAUNCH(false, 4, 4, 32, 2, 2);
    else if (conv_id == 3) WLAUNCH(false, 3, 3, 64, 1, 2);
    else TORCH_CHECK(false, "unknown conv_id");
#undef WLAUNCH
    return {dWt, db};
}
