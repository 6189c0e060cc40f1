// Fused persistent LSTM kernels (gfx950) — SURVEY.md §2.3 K6, the hard one.
//
// Replaces nn.LSTM + pack_padded_sequence (reference model.py:95-100,136-141;
// 87k hipBLASLt launches per profile) with ONE kernel per direction.
//
// Structure (forward):
// - X = rin @ W_ih^T + b precomputed for all (B, T) steps by the MFMA GEMM.
// - Persistent launch, one workgroup per 16 hidden units x batch half
//   (x2 networks in one launch = 128 workgroups; slice width and batch
//   split are template parameters — 8-unit and quartered variants stay
//   selectable for comparison).  W_hh slice and the cell state c stay
//   LDS-resident for all T steps; each step bulk-stages the previous h
//   into LDS with coalesced 16 B loads, runs the gate GEMM entirely out
//   of LDS, and advances h/c with vectorized stores.
// - NO global barrier: the time index makes the h exchange naturally
//   double-buffered, so each step is synchronized by PRODUCER FLAGS only
//   (HIP guide §6 G16: plain h stores -> per-wave vmcnt drain ->
//   __syncthreads -> release fence -> relaxed per-slice flag; consumers
//   poll 64 flags lane-parallel, one acquire fence, then plain loads).
//   Flag words are zeroed by a tiny kernel before every launch (a
//   CAPTURED hipMemsetAsync replays a garbage fill value — see
//   zero_gridbar_kernel); max skew
//   between workgroups is self-limiting (each wg is both producer and
//   consumer).  Spins are bounded; a poison word aborts the launch.
// - Per-sample length masks replace pack_padded semantics: masked steps
//   copy h/c through unchanged.
//
// Backward (online net): reverse recurrence with the same counter protocol
// on the dgates stream — one workgroup per 16 hidden units x batch QUARTER
// (128 workgroups; measured faster than batch halves — the backward is
// CU-coverage bound); the recurrent GEMM streams dgates_{t+1} through a
// 4-deep register prefetch ring with the 4H reduction split across waves.
// The bulk weight gradients (dW_hh, dW_ih, db) and dX are plain GEMMs
// outside the kernel (gemm_wgrad / gemm_dgrad over B*T rows).
//
// Gate order matches torch.nn.LSTM: [i, f, g, o] chunks of H.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

union lbf8u {
    bf16x8 v;
    uint4 u;
};

__device__ __forceinline__ bf16x8 lload8(const __hip_bfloat16* p) {
    lbf8u r;
    r.u = *reinterpret_cast<const uint4*>(p);
    return r.v;
}

__device__ __forceinline__ void lstore8(__hip_bfloat16* p, bf16x8 v) {
    lbf8u r;
    r.v = v;
    *reinterpret_cast<uint4*>(p) = r.u;
}

__device__ __forceinline__ bf16x8 lzero8() {
    lbf8u r;
    r.u = uint4{0, 0, 0, 0};
    return r.v;
}

__device__ __forceinline__ float sigmoidf_(float x) {
    return 1.f / (1.f + __expf(-x));
}

// ---------------------------------------------------------------------------
// Producer-flag hand-off.  flags[slice] counts published time indices of that
// producer slice.  publish: called by ALL threads after the slice stores.
// await: wave 0 polls `nflags` flag words lane-parallel (relaxed), one
// acquire fence, __syncthreads.  Returns false on bounded-spin timeout.
// ---------------------------------------------------------------------------
#define LSTM_MAX_FLAGS 256

__device__ __forceinline__ void publish_slice(unsigned* flag, unsigned value) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every storing wave
    __syncthreads();
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_store(flag, value, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
    }
}

__device__ __forceinline__ bool await_slices(unsigned* flags, int nflags,
                                             unsigned value, unsigned* poison) {
    __shared__ unsigned ok_s;
    if (threadIdx.x < WAVE) {
        int lane = threadIdx.x;
        unsigned ok = 1;
        long spins = 0;
        for (;;) {
            unsigned f1 = (lane < nflags)
                ? __hip_atomic_load(&flags[lane], __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT) : ~0u;
            unsigned f2 = (lane + WAVE < nflags)
                ? __hip_atomic_load(&flags[lane + WAVE], __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT) : ~0u;
            if (__all(f1 >= value && f2 >= value)) break;
            __builtin_amdgcn_s_sleep(1);
            if (++spins > (long)2e8) {
                if (lane == 0)
                    __hip_atomic_store(poison, 1u, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                ok = 0;
                break;
            }
            if (__hip_atomic_load(poison, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT)) {
                ok = 0;
                break;
            }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        if (lane == 0) ok_s = ok;
    }
    __syncthreads();
    return ok_s != 0;
}

// ---------------------------------------------------------------------------
// Counter-aggregated hand-off (the production protocol): ONE atomic counter
// per (net, batch-half) group on its own cacheline.  publish = release
// fence + atomicAdd(1); await = poll until counter >= nprod * value.  The
// protocol's self-limiting skew (a producer cannot be >1 step ahead of the
// slowest in its group — it must await the whole group before advancing)
// makes the sum threshold equivalent to per-producer flags, at ~2/3 the
// per-step cost (tools/lstm_handoff_bench.py: 2.4 vs 3.4 us at 64 wgs).
// ---------------------------------------------------------------------------
__device__ __forceinline__ void publish_count(unsigned* ctr) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every storing wave
    __syncthreads();
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_fetch_add(ctr, 1u, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
    }
}

__device__ __forceinline__ bool await_count(unsigned* ctr, unsigned target,
                                            unsigned* poison) {
    __shared__ unsigned ok_c;
    if (threadIdx.x == 0) {
        unsigned ok = 1;
        long spins = 0;
        while (__hip_atomic_load(ctr, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) < target) {
            __builtin_amdgcn_s_sleep(1);
            if (++spins > (long)2e8) {
                __hip_atomic_store(poison, 1u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
                ok = 0;
                break;
            }
            if (__hip_atomic_load(poison, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT)) {
                ok = 0;
                break;
            }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        ok_c = ok;
    }
    __syncthreads();
    return ok_c != 0;
}

// layout of the workspace (int32 words): [0..255] counters/flags (group g
// uses word g*32 — one cacheline apart), [256] poison, then legacy barrier
// words
struct GridBar {
    unsigned flags[LSTM_MAX_FLAGS];
    unsigned poison;
    unsigned bucket[8];
    unsigned top;
    unsigned gen;
    unsigned gen_b[8];
};

// legacy counter barrier, kept for the microbench / comparison
__device__ __forceinline__ bool grid_barrier(GridBar* bar, unsigned epoch,
                                             int nblocks) {
    __syncthreads();
    __shared__ unsigned ok_s;
    if (threadIdx.x == 0) {
        int b = blockIdx.x & 7;
        int per = (nblocks + 7 - b) >> 3;
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        unsigned prev = __hip_atomic_fetch_add(&bar->bucket[b], 1u,
                                               __ATOMIC_RELAXED,
                                               __HIP_MEMORY_SCOPE_AGENT);
        if (prev == epoch * (unsigned)per - 1u) {
            unsigned p2 = __hip_atomic_fetch_add(&bar->top, 1u,
                                                 __ATOMIC_RELAXED,
                                                 __HIP_MEMORY_SCOPE_AGENT);
            if (p2 == epoch * 8u - 1u)
                __hip_atomic_store(&bar->gen, epoch, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
        }
        unsigned ok = 1;
        long spins = 0;
        while (__hip_atomic_load(&bar->gen, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) < epoch) {
            __builtin_amdgcn_s_sleep(2);
            if (++spins > (long)2e8) { ok = 0; break; }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        ok_s = ok;
    }
    __syncthreads();
    return ok_s != 0;
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
// BROWS: batch rows per workgroup.  64 = classic layout (whole batch).
// 32 = batch-split: twice the workgroups, half the LDS staging per step,
// flag groups per (net, half).
// UNITS: hidden units per workgroup.  8 = classic; 16 halves the workgroup
// count (half the handoff producers per counter group AND half the total
// h re-read traffic) at twice the per-wg MFMA work — the per-step cost is
// latency-dominated, so fewer/fatter workgroups win (measured; VERDICT r1
// item 7).
template <int H, int BROWS = 64, int UNITS = 8>
__global__ __launch_bounds__(256, 1) void lstm_fwd_kernel(
    const __hip_bfloat16* __restrict__ X0,    // (B, T, 4H)
    const __hip_bfloat16* __restrict__ X1,    // or null
    const __hip_bfloat16* __restrict__ Whh0,  // (4H, H) row-major
    const __hip_bfloat16* __restrict__ Whh1,
    const float* __restrict__ init0,          // (2, B, H): h0, c0
    const float* __restrict__ init1,
    const int* __restrict__ lens,             // (B,)
    __hip_bfloat16* __restrict__ Hout0,       // (B, T+1, H)
    __hip_bfloat16* __restrict__ Hout1,
    float* __restrict__ Cout0,                // (B, T+1, H)
    float* __restrict__ Cout1,
    __hip_bfloat16* __restrict__ stash0,      // (B, T, 4H) post-nonlin gates
    GridBar* bar, int B, int T, int nblocks, int nhalves) {
    constexpr int WGS_PER_HALF = H / UNITS;
    constexpr int GCOLS = 4 * UNITS;          // gate columns per wg
    const int wgs_per_net = WGS_PER_HALF * nhalves;
    const int net = blockIdx.x / wgs_per_net;
    const int rem = blockIdx.x % wgs_per_net;
    const int half = rem / WGS_PER_HALF;
    const int wid = rem % WGS_PER_HALF;
    const int u0 = wid * UNITS;
    const int b0 = half * BROWS;
    const int Bl = min(B - b0, BROWS);

    const __hip_bfloat16* X = net ? X1 : X0;
    const __hip_bfloat16* Whh = net ? Whh1 : Whh0;
    const float* init = net ? init1 : init0;
    __hip_bfloat16* Hout = net ? Hout1 : Hout0;
    float* Cout = net ? Cout1 : Cout0;
    __hip_bfloat16* stash = net ? nullptr : stash0;
    unsigned* ctr = &bar->flags[(net * nhalves + half) * 32];

    __shared__ __hip_bfloat16 s_whh[GCOLS][H + 8];
    __shared__ __hip_bfloat16 s_h[BROWS][H + 8];
    __shared__ float s_gates[BROWS][GCOLS + 4];
    __shared__ float s_c[BROWS][UNITS];
    __shared__ __hip_bfloat16 s_hrow[BROWS][UNITS];

    for (int e = threadIdx.x * 8; e < GCOLS * H; e += blockDim.x * 8) {
        int c = e / H;
        int k = e % H;
        int g = c / UNITS, j = c % UNITS;
        lstore8(&s_whh[c][k], lload8(Whh + (long)(g * H + u0 + j) * H + k));
    }
    // h0 -> Hout[:,0] (this wg's slice); c0 -> LDS-resident cell state
    for (int p = threadIdx.x; p < Bl * UNITS; p += blockDim.x) {
        int bl = p / UNITS, j = p % UNITS;
        int b = b0 + bl;
        int u = u0 + j;
        Hout[((long)b * (T + 1)) * H + u] = f2bf(init[(long)b * H + u]);
        Cout[((long)b * (T + 1)) * H + u] = init[((long)B + b) * H + u];
        s_c[bl][j] = init[((long)B + b) * H + u];
    }
    publish_count(ctr);

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    // wave tiling over (BROWS rows x GCOLS cols):
    //   GCOLS=32: 2x2 waves of 16x16xNFRAG; GCOLS=64: 4 col-waves, rows
    //   split into BROWS/16 fragments per wave.
    constexpr int NCW = GCOLS / 16;            // col fragments total
    constexpr int NFRAG = (NCW == 4) ? BROWS / 16 : BROWS / 32;
    const int wrow0 = (NCW == 4) ? 0 : (wave >> 1) * 16 * NFRAG;
    const int wcol0 = (NCW == 4) ? wave * 16 : (wave & 1) * 16;
    const int frow = lane & 15;
    const int kseg = (lane >> 4) * 8;

    constexpr int CHUNKS = (BROWS * H) / 8;

    for (int t = 0; t < T; ++t) {
        if (!await_count(ctr, (unsigned)WGS_PER_HALF * (t + 1),
                         &bar->poison))
            return;
        // bulk-stage this half's h_prev (Bl x H) into LDS
        {
            const long base = (long)t * H;
#pragma unroll
            for (int e = threadIdx.x; e < CHUNKS; e += 256) {
                int row = e / (H / 8);
                int k8 = (e % (H / 8)) * 8;
                bf16x8 v = (row < Bl)
                    ? lload8(Hout + ((long)(b0 + row) * (T + 1)) * H + base + k8)
                    : lzero8();
                lstore8(&s_h[row][k8], v);
            }
        }
        __syncthreads();

        // gates = h_prev @ Whh_slice^T  (all-LDS MFMA)
        f32x4 acc[NFRAG] = {};
#pragma unroll
        for (int k0 = 0; k0 < H; k0 += 32) {
            bf16x8 bfr = lload8(&s_whh[wcol0 + frow][k0 + kseg]);
#pragma unroll
            for (int i = 0; i < NFRAG; ++i) {
                bf16x8 afr = lload8(&s_h[wrow0 + i * 16 + frow][k0 + kseg]);
                acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr, bfr, acc[i], 0, 0, 0);
            }
        }
        {
            int ccol = lane & 15;
            int crow = (lane >> 4) * 4;
#pragma unroll
            for (int i = 0; i < NFRAG; ++i)
#pragma unroll
                for (int r = 0; r < 4; ++r)
                    s_gates[wrow0 + i * 16 + crow + r][wcol0 + ccol] = acc[i][r];
        }
        __syncthreads();
        // + X[t] (vectorized), then nonlinearities + state advance
        {
            int row = threadIdx.x / 4;
            int g = threadIdx.x % 4;
            if (row < Bl) {
#pragma unroll
                for (int j8 = 0; j8 < UNITS; j8 += 8) {
                    bf16x8 x8 = lload8(
                        X + ((long)(b0 + row) * T + t) * 4 * H + g * H + u0
                        + j8);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        s_gates[row][g * UNITS + j8 + j] += (float)x8[j];
                }
            }
        }
        __syncthreads();
        for (int p = threadIdx.x; p < Bl * UNITS; p += blockDim.x) {
            int b = p / UNITS, j = p % UNITS;
            bool active = t < lens[b0 + b];
            float i_ = 0.f, f_ = 0.f, g_ = 0.f, o_ = 0.f;
            float c = s_c[b][j], h;
            if (active) {
                i_ = sigmoidf_(s_gates[b][0 * UNITS + j]);
                f_ = sigmoidf_(s_gates[b][1 * UNITS + j]);
                g_ = tanhf(s_gates[b][2 * UNITS + j]);
                o_ = sigmoidf_(s_gates[b][3 * UNITS + j]);
                c = f_ * c + i_ * g_;
                h = o_ * tanhf(c);
            } else {
                h = (float)*(const __bf16*)&s_h[b][u0 + j];
            }
            s_c[b][j] = c;
            *(__bf16*)&s_hrow[b][j] = (__bf16)h;
            s_gates[b][0 * UNITS + j] = i_;
            s_gates[b][1 * UNITS + j] = f_;
            s_gates[b][2 * UNITS + j] = g_;
            s_gates[b][3 * UNITS + j] = o_;
        }
        __syncthreads();
        // vectorized writers: h slice, c, gate stash
        {
            int tid = threadIdx.x;
            constexpr int H8 = UNITS / 8;        // 16-B pieces per h row
            if (tid < 64 * H8) {
                int b = tid / H8, piece = tid % H8;
                if (b < Bl) {
                    long off = ((long)(b0 + b) * (T + 1) + t + 1) * H + u0
                               + piece * 8;
                    lstore8(Hout + off, *reinterpret_cast<bf16x8*>(
                        &s_hrow[b][piece * 8]));
                }
            } else if (tid >= 128) {
                // c writers: threads [128, 256) cover BROWS rows x C4
                // float4 pieces (BROWS*C4 <= 128 for both layouts)
                int t2 = tid - 128;
                constexpr int C4 = UNITS / 4;    // float4 pieces per c row
                static_assert(BROWS * C4 <= 128, "c-writer thread budget");
                int b = t2 / C4, ch = t2 % C4;
                if (b < Bl) {
                    long off = ((long)(b0 + b) * (T + 1) + t + 1) * H + u0
                               + ch * 4;
                    *reinterpret_cast<float4*>(Cout + off) =
                        *reinterpret_cast<float4*>(&s_c[b][ch * 4]);
                }
            }
        }
        if (stash) {
            int row = threadIdx.x / 4;
            int g = threadIdx.x % 4;
            if (row < Bl) {
#pragma unroll
                for (int j8 = 0; j8 < UNITS; j8 += 8) {
                    bf16x8 v;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        v[j] = (__bf16)s_gates[row][g * UNITS + j8 + j];
                    lstore8(stash + ((long)(b0 + row) * T + t) * 4 * H
                            + g * H + u0 + j8, v);
                }
            }
        }
        publish_count(ctr);
    }
}

// ---------------------------------------------------------------------------
// Backward (online net).  H/16 wgs, each 16 hidden units; W_hh^T slice in
// LDS; dgates_{t+1} streamed through LDS in 256-col pieces; vectorized row
// staging of stash/C/dHext; flag hand-off on the dgates stream.
// ---------------------------------------------------------------------------
// BROWS as in lstm_fwd_kernel: 32 doubles the workgroup count (batch
// halves with their own flag groups) and splits the recurrent GEMM's K
// reduction across wave pairs.  UNITS=32 (with BROWS=32) halves the
// workgroup count again — fewer/fatter workgroups, half the handoff
// producers per counter, waves tile (row half x col half) at full K
// (s_wb then holds 32 u-rows x 4H ~ 131 KB of the 160 KB LDS).
template <int H, int BROWS = 64, int UNITS = 16>
__global__ __launch_bounds__(256, 1) void lstm_bwd_kernel(
    const __hip_bfloat16* __restrict__ stash,  // (B, T, 4H) i,f,g,o
    const float* __restrict__ Cout,            // (B, T+1, H)
    const __hip_bfloat16* __restrict__ Hout,   // (unused)
    const float* __restrict__ dHext,           // (B, T, H) upstream
    const __hip_bfloat16* __restrict__ Whh_bwd,// (H, 4H): W_hh^T row-major
    const int* __restrict__ lens,
    __hip_bfloat16* __restrict__ dgates,       // (B, T, 4H) out
    GridBar* bar, int B, int T, int nblocks, int nhalves) {
    constexpr int WGS = H / UNITS;
    constexpr int KSPLIT = (UNITS == 16 && BROWS == 32) ? 2
                           : (UNITS == 16 && BROWS <= 16) ? 4 : 1;
    constexpr int NCWAVE = UNITS / 16;         // col fragments across waves
    const int half = blockIdx.x / WGS;
    const int wid = blockIdx.x % WGS;
    const int u0 = wid * UNITS;
    const int b0 = half * BROWS;
    const int Bl = min(B - b0, BROWS);

    __shared__ __hip_bfloat16 s_wb[UNITS][4 * H + 8];
    __shared__ float s_dh[BROWS][UNITS + 1];
    __shared__ float s_dc[BROWS][UNITS + 1];
    __shared__ float s_rec[KSPLIT][BROWS][UNITS + 1];
    __shared__ __hip_bfloat16 s_dgout[BROWS][4 * UNITS + 8];

    unsigned* ctr = &bar->flags[half * 32];

    for (int e = threadIdx.x * 8; e < UNITS * 4 * H; e += blockDim.x * 8) {
        int c = e / (4 * H);
        int k = e % (4 * H);
        lstore8(&s_wb[c][k], lload8(Whh_bwd + (long)(u0 + c) * 4 * H + k));
    }
    for (int p = threadIdx.x; p < BROWS * UNITS; p += blockDim.x) {
        s_dh[p / UNITS][p % UNITS] = 0.f;
        s_dc[p / UNITS][p % UNITS] = 0.f;
    }
    publish_count(ctr);

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    // UNITS=16, KSPLIT=1: 4 waves x 16-row tiles, full-K.
    // UNITS=16, KSPLIT=2: wave pairs split the 4H reduction.
    // UNITS=32: waves tile (row half x col half), full-K.
    const int wrow0 = (NCWAVE == 2) ? (wave >> 1) * 16
                      : (KSPLIT == 4) ? 0
                      : (KSPLIT == 1 ? wave : (wave & 1)) * 16;
    const int wcol0 = (NCWAVE == 2) ? (wave & 1) * 16 : 0;
    const int wk = (KSPLIT == 1) ? 0 : (KSPLIT == 4) ? wave : (wave >> 1);
    const int frow = lane & 15;
    const int kseg = (lane >> 4) * 8;
    constexpr int NCHUNK = 4 * H / 32;
    const int c0 = wk * (NCHUNK / KSPLIT);
    const int cN = c0 + NCHUNK / KSPLIT;

    for (int t = T - 1; t >= 0; --t) {
        unsigned need = (unsigned)(T - t);   // pieces published for t+1
        if (!await_count(ctr, (unsigned)WGS * need, &bar->poison)) return;
        // rec(B,UNITS) = dgates_{t+1}(B,4H) @ s_wb^T with a 4-deep register
        // prefetch ring on the dgates stream (loads stay in flight across
        // MFMAs; no per-piece barriers)
        if (t < T - 1) {
            const int arow_l = wrow0 + frow;
            const __hip_bfloat16* dgrow =
                dgates + ((long)(b0 + arow_l) * T + t + 1) * 4 * H;
            const bool rvalid = arow_l < Bl;
            f32x4 acc = {};
            bf16x8 a0 = rvalid ? lload8(dgrow + (c0 + 0) * 32 + kseg) : lzero8();
            bf16x8 a1 = rvalid ? lload8(dgrow + (c0 + 1) * 32 + kseg) : lzero8();
            bf16x8 a2 = rvalid ? lload8(dgrow + (c0 + 2) * 32 + kseg) : lzero8();
            bf16x8 a3 = rvalid ? lload8(dgrow + (c0 + 3) * 32 + kseg) : lzero8();
#pragma unroll 4
            for (int kc = c0; kc < cN; kc += 4) {
                bf16x8 b0 = lload8(&s_wb[wcol0 + frow][(kc + 0) * 32 + kseg]);
                bf16x8 b1 = lload8(&s_wb[wcol0 + frow][(kc + 1) * 32 + kseg]);
                bf16x8 b2 = lload8(&s_wb[wcol0 + frow][(kc + 2) * 32 + kseg]);
                bf16x8 b3 = lload8(&s_wb[wcol0 + frow][(kc + 3) * 32 + kseg]);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
                if (kc + 4 < cN)
                    a0 = rvalid ? lload8(dgrow + (kc + 4) * 32 + kseg) : lzero8();
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
                if (kc + 5 < cN)
                    a1 = rvalid ? lload8(dgrow + (kc + 5) * 32 + kseg) : lzero8();
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc, 0, 0, 0);
                if (kc + 6 < cN)
                    a2 = rvalid ? lload8(dgrow + (kc + 6) * 32 + kseg) : lzero8();
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc, 0, 0, 0);
                if (kc + 7 < cN)
                    a3 = rvalid ? lload8(dgrow + (kc + 7) * 32 + kseg) : lzero8();
            }
            int ccol = lane & 15;
            int crow = (lane >> 4) * 4;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = wrow0 + crow + r;
                if (row < BROWS) s_rec[wk][row][wcol0 + ccol] = acc[r];
            }
        }
        __syncthreads();
        for (int p = threadIdx.x; p < Bl * UNITS; p += blockDim.x) {
            int b = p / UNITS, jl = p % UNITS;
            int bg = b0 + b;
            int u = u0 + jl;
            float dh, dc_in;
            bool last = (t == T - 1);
            bool active_next = !last && ((t + 1) < lens[bg]);
            float ext = dHext[((long)bg * T + t) * H + u];
            float rec = s_rec[0][b][jl];
#pragma unroll
            for (int kk = 1; kk < KSPLIT; ++kk) rec += s_rec[kk][b][jl];
            if (last) {
                dh = ext;
                dc_in = 0.f;
            } else if (active_next) {
                long so1 = ((long)bg * T + t + 1) * 4 * H + u;
                float f_next = bf2f(stash[so1 + H]);
                dh = ext + rec;
                dc_in = s_dc[b][jl] * f_next;
            } else {
                dh = ext + s_dh[b][jl];
                dc_in = s_dc[b][jl];
            }
            bool active = t < lens[bg];
            long so = ((long)bg * T + t) * 4 * H + u;
            float di = 0.f, df = 0.f, dg = 0.f, do_ = 0.f;
            if (active) {
                float i_ = bf2f(stash[so]);
                float f_ = bf2f(stash[so + H]);
                float g_ = bf2f(stash[so + 2 * H]);
                float o_ = bf2f(stash[so + 3 * H]);
                float tc = tanhf(Cout[((long)bg * (T + 1) + t + 1) * H + u]);
                float c_prev = Cout[((long)bg * (T + 1) + t) * H + u];
                float dc = dc_in + dh * o_ * (1.f - tc * tc);
                di = dc * g_ * i_ * (1.f - i_);
                df = dc * c_prev * f_ * (1.f - f_);
                dg = dc * i_ * (1.f - g_ * g_);
                do_ = dh * tc * o_ * (1.f - o_);
                s_dh[b][jl] = dh;
                s_dc[b][jl] = dc;
            } else {
                s_dh[b][jl] = dh;
                s_dc[b][jl] = dc_in;
            }
            *(__bf16*)&s_dgout[b][0 * UNITS + jl] = (__bf16)di;
            *(__bf16*)&s_dgout[b][1 * UNITS + jl] = (__bf16)df;
            *(__bf16*)&s_dgout[b][2 * UNITS + jl] = (__bf16)dg;
            *(__bf16*)&s_dgout[b][3 * UNITS + jl] = (__bf16)do_;
        }
        __syncthreads();
        // vectorized dgates writes: thread (b, g) -> UNITS/8 x 16 B
        {
            int tid = threadIdx.x;
            int b = tid / 4;
            int g = tid % 4;
            if (b < Bl) {
#pragma unroll
                for (int rep = 0; rep < UNITS / 8; ++rep) {
                    long off = ((long)(b0 + b) * T + t) * 4 * H + g * H + u0
                               + rep * 8;
                    lstore8(dgates + off,
                            *reinterpret_cast<bf16x8*>(
                                &s_dgout[b][g * UNITS + rep * 8]));
                }
            }
        }
        publish_count(ctr);
    }
}

// counter-aggregated handoff microbench: ONE atomic counter per group —
// publish = single atomicAdd, await = poll one word until nblocks*(t+1).
__global__ void handoff_counter_bench_kernel(GridBar* bar, int steps,
                                             int nblocks) {
    unsigned* ctr = &bar->flags[0];
    for (int t = 0; t < steps; ++t) {
        __syncthreads();
        if (threadIdx.x == 0) {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
            __hip_atomic_fetch_add(ctr, 1u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
        }
        if (threadIdx.x == 0) {
            long spins = 0;
            while (__hip_atomic_load(ctr, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT)
                   < (unsigned)(nblocks * (t + 1))) {
                __builtin_amdgcn_s_sleep(1);
                if (++spins > (long)2e8) return;
            }
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// assemble_rin: build the padded LSTM input rows (latent | one-hot action |
// reward | zero pad) in ONE launch (replaces a zeros fill + three slice
// copies per network per step).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void assemble_rin_kernel(
    const __hip_bfloat16* __restrict__ latent,  // (M, 512)
    const float* __restrict__ la,               // (M, A)
    const float* __restrict__ lr,               // (M,)
    __hip_bfloat16* __restrict__ rin,           // (M, KP)
    int M, int A, int KP) {
    unsigned idx = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (idx >= (unsigned)M * KP) return;
    int k = (int)(idx % (unsigned)KP);
    unsigned m = idx / (unsigned)KP;
    if (k + 8 <= 512) {
        *reinterpret_cast<bf16x8*>(rin + idx) =
            lload8(latent + (long)m * 512 + k);
        return;
    }
    bf16x8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        int kk = k + e;
        float f = 0.f;
        if (kk < 512) f = bf2f(latent[(long)m * 512 + kk]);
        else if (kk < 512 + A) f = la[(long)m * A + (kk - 512)];
        else if (kk == 512 + A) f = lr[m];
        v[e] = (__bf16)f;
    }
    *reinterpret_cast<bf16x8*>(rin + idx) = v;
}

// barrier-only microbench
__global__ void barrier_bench_kernel(GridBar* bar, int steps, int nblocks) {
    for (int t = 0; t < steps; ++t)
        if (!grid_barrier(bar, (unsigned)(t + 1), nblocks)) return;
}

// handoff-only microbench: prices one flag publish/await round at `nblocks`
__global__ void handoff_bench_kernel(GridBar* bar, int steps, int nblocks) {
    unsigned* flags = bar->flags;
    unsigned* myflag = &flags[blockIdx.x];
    if (threadIdx.x == 0)
        __hip_atomic_store(myflag, 1u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
    for (int t = 0; t < steps; ++t) {
        if (!await_slices(flags, nblocks, (unsigned)(t + 1), &bar->poison))
            return;
        publish_slice(myflag, (unsigned)(t + 2));
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

// NOT hipMemsetAsync: a captured memset node replays with a garbage fill
// value from the second hipGraph replay onward (observed on ROCm 7.2,
// gpurun_out/graph_probe2.log — counters came back as <garbage>+increments
// and the poison word as the raw garbage).  A plain kernel replays
// correctly.
__global__ void zero_gridbar_kernel(unsigned* ws) {
    if (threadIdx.x < sizeof(GridBar) / 4) ws[threadIdx.x] = 0u;
}

static void zero_ws(torch::Tensor& ws, hipStream_t stream) {
    TORCH_CHECK(ws.numel() * ws.element_size() >= (long)sizeof(GridBar),
                "barrier workspace too small (need >= 256 int32)");
    hipLaunchKernelGGL(zero_gridbar_kernel, dim3(1), dim3(320), 0, stream,
                       reinterpret_cast<unsigned*>(ws.data_ptr()));
}

torch::Tensor assemble_rin(torch::Tensor latent, torch::Tensor la,
                           torch::Tensor lr, int64_t kin_pad) {
    long M = latent.size(0);
    long A = la.size(1);
    TORCH_CHECK(latent.dtype() == torch::kBFloat16 && latent.is_contiguous());
    TORCH_CHECK(la.dtype() == torch::kFloat32 && la.is_contiguous());
    TORCH_CHECK(kin_pad % 32 == 0 && 512 + A + 1 <= kin_pad);
    auto rin = torch::empty({M, kin_pad}, latent.options());
    long total = M * kin_pad / 8;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(assemble_rin_kernel,
                       dim3((int)((total + 255) / 256)), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(latent.data_ptr()),
                       la.data_ptr<float>(), lr.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(rin.data_ptr()),
                       (int)M, (int)A, (int)kin_pad);
    return rin;
}

void barrier_bench(torch::Tensor barrier_ws, int64_t steps, int64_t nblocks) {
    auto stream = at::cuda::getCurrentCUDAStream();
    zero_ws(barrier_ws, stream.stream());
    hipLaunchKernelGGL(barrier_bench_kernel, dim3((int)nblocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),
                       (int)steps, (int)nblocks);
}

void handoff_counter_bench(torch::Tensor barrier_ws, int64_t steps,
                           int64_t nblocks) {
    auto stream = at::cuda::getCurrentCUDAStream();
    zero_ws(barrier_ws, stream.stream());
    hipLaunchKernelGGL(handoff_counter_bench_kernel, dim3((int)nblocks),
                       dim3(256), 0, stream.stream(),
                       reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),
                       (int)steps, (int)nblocks);
}

void handoff_bench(torch::Tensor barrier_ws, int64_t steps, int64_t nblocks) {
    auto stream = at::cuda::getCurrentCUDAStream();
    zero_ws(barrier_ws, stream.stream());
    hipLaunchKernelGGL(handoff_bench_kernel, dim3((int)nblocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),
                       (int)steps, (int)nblocks);
}

std::vector<torch::Tensor> lstm_fwd(
    torch::Tensor X0, torch::Tensor X1, torch::Tensor Whh0, torch::Tensor Whh1,
    torch::Tensor init0, torch::Tensor init1, torch::Tensor lens,
    torch::Tensor barrier_ws, bool want_stash) {
    TORCH_CHECK(X0.is_cuda() && X0.dtype() == torch::kBFloat16 && X0.is_contiguous());
    long B = X0.size(0), T = X0.size(1);
    long H4 = X0.size(2);
    long H = H4 / 4;
    TORCH_CHECK(H == 512, "lstm_fwd is instantiated for H=512");
    TORCH_CHECK(B <= 64, "B <= 64 per launch");
    bool two = X1.defined() && X1.numel() > 0;

    auto bf = X0.options();
    auto f32 = X0.options().dtype(torch::kFloat32);
    auto H0 = torch::empty({B, T + 1, H}, bf);
    auto C0 = torch::empty({B, T + 1, H}, f32);
    auto H1 = two ? torch::empty({B, T + 1, H}, bf) : torch::Tensor();
    auto C1 = two ? torch::empty({B, T + 1, H}, f32) : torch::Tensor();
    auto stash = want_stash ? torch::empty({B, T, H4}, bf) : torch::Tensor();

    // UNITS: hidden units per workgroup.  16 is the measured default for
    // the dual-net batch-split launch (fewer/fatter workgroups: half the
    // handoff producers per counter and half the total h re-read);
    // R2D2_LSTM_UNITS=8 selects the classic layout for comparison.
    static const int units_env = [] {
        const char* e = getenv("R2D2_LSTM_UNITS");
        return e ? atoi(e) : 16;
    }();
    // R2D2_LSTM_FWD_BROWS=16 quarters the batch (256 wgs for the dual-net
    // launch = every CU)
    static const int fbrows_env = [] {
        const char* e = getenv("R2D2_LSTM_FWD_BROWS");
        return e ? atoi(e) : 32;
    }();
    int units = (units_env == 8) ? 8 : 16;
    int nhalves = B > 32 ? 2 : 1;
    if (nhalves == 1) units = 8;   // single-half keeps the classic layout
    int fbrows = (nhalves == 2 && units == 16 && fbrows_env == 16 && B > 48)
                     ? 16 : (nhalves == 2 ? 32 : 64);
    if (fbrows == 16) nhalves = 4;
    int wgs = (int)H / units;
    int nblocks = wgs * (two ? 2 : 1) * nhalves;
    auto stream = at::cuda::getCurrentCUDAStream();
    zero_ws(barrier_ws, stream.stream());

    auto bp = [](torch::Tensor& t) {
        return t.defined()
            ? reinterpret_cast<__hip_bfloat16*>(t.data_ptr()) : nullptr;
    };
#define LSTMF(BROWS_, UNITS_)                                                 \
    hipLaunchKernelGGL((lstm_fwd_kernel<512, BROWS_, UNITS_>), dim3(nblocks), \
        dim3(256), 0, stream.stream(),                                        \
        reinterpret_cast<const __hip_bfloat16*>(X0.data_ptr()),               \
        two ? reinterpret_cast<const __hip_bfloat16*>(X1.data_ptr()) : nullptr,\
        reinterpret_cast<const __hip_bfloat16*>(Whh0.data_ptr()),             \
        two ? reinterpret_cast<const __hip_bfloat16*>(Whh1.data_ptr()) : nullptr,\
        init0.data_ptr<float>(),                                              \
        two ? init1.data_ptr<float>() : nullptr,                              \
        lens.data_ptr<int>(), bp(H0), bp(H1),                                 \
        C0.data_ptr<float>(), two ? C1.data_ptr<float>() : nullptr,           \
        want_stash ? bp(stash) : nullptr,                                     \
        reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),                    \
        (int)B, (int)T, nblocks, nhalves)
    if (fbrows == 16) LSTMF(16, 16);
    else if (nhalves == 2) {
        if (units == 16) LSTMF(32, 16); else LSTMF(32, 8);
    } else {
        LSTMF(64, 8);
    }
#undef LSTMF

    std::vector<torch::Tensor> out = {H0, C0};
    out.push_back(two ? H1 : torch::Tensor());
    out.push_back(two ? C1 : torch::Tensor());
    out.push_back(want_stash ? stash : torch::Tensor());
    return out;
}

torch::Tensor lstm_bwd(torch::Tensor stash, torch::Tensor Cout,
                       torch::Tensor Hout, torch::Tensor dHext,
                       torch::Tensor Whh_bwd, torch::Tensor lens,
                       torch::Tensor barrier_ws) {
    long B = stash.size(0), T = stash.size(1), H4 = stash.size(2);
    long H = H4 / 4;
    TORCH_CHECK(H == 512, "lstm_bwd is instantiated for H=512");
    auto dgates = torch::empty({B, T, H4}, stash.options());
    // Measured DEAD END (docs/KERNELS.md): 32-unit bwd slices
    // (R2D2_LSTM_BWD_UNITS=32) run 1.34 ms vs 0.94 ms at 16 — the 131 KB
    // s_wb slab and the narrower 16-row dgates A-tiles at full K=2048 cost
    // more than the halved handoff saves.  16 stays the default (the fwd
    // kernel, whose weight slab is 4x smaller, DOES win at 16 units).
    static const int bunits_env = [] {
        const char* e = getenv("R2D2_LSTM_BWD_UNITS");
        return e ? atoi(e) : 16;
    }();
    // batch QUARTERED by default (128 wgs, KSPLIT=4): measured 0.71 vs
    // 0.95 ms at BROWS=32 (the backward is CU-coverage bound).  Pushing
    // further to BROWS=8 (256 wgs) LOSES (12.03k vs 12.43k end-to-end) —
    // 8-row batches leave the 16-row MFMA A-fragments half empty.
    // R2D2_LSTM_BWD_BROWS selects 8/16/32 for comparison.
    static const int brows_env = [] {
        const char* e = getenv("R2D2_LSTM_BWD_BROWS");
        return e ? atoi(e) : 16;
    }();
    int nhalves = B > 32 ? 2 : 1;
    int units = (nhalves == 2 && bunits_env == 32) ? 32 : 16;
    int brows = (nhalves == 2 && units == 16 && B > 48)
                    ? brows_env : (nhalves == 2 ? 32 : 64);
    if (brows != 8 && brows != 16 && brows != 32) brows = 16;
    if (brows == 16) nhalves = 4;
    else if (brows == 8) nhalves = 8;
    int nblocks = (int)H / units * nhalves;
    auto stream = at::cuda::getCurrentCUDAStream();
    zero_ws(barrier_ws, stream.stream());
#define LSTMB(BROWS_, UNITS_)                                                 \
    hipLaunchKernelGGL((lstm_bwd_kernel<512, BROWS_, UNITS_>), dim3(nblocks), \
        dim3(256), 0, stream.stream(),                                        \
        reinterpret_cast<const __hip_bfloat16*>(stash.data_ptr()),            \
        Cout.data_ptr<float>(),                                               \
        reinterpret_cast<const __hip_bfloat16*>(Hout.data_ptr()),             \
        dHext.data_ptr<float>(),                                              \
        reinterpret_cast<const __hip_bfloat16*>(Whh_bwd.data_ptr()),          \
        lens.data_ptr<int>(),                                                 \
        reinterpret_cast<__hip_bfloat16*>(dgates.data_ptr()),                 \
        reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),                    \
        (int)B, (int)T, nblocks, nhalves)
    if (brows == 8) LSTMB(8, 16);
    else if (brows == 16) LSTMB(16, 16);
    else if (nhalves == 2) {
        if (units == 32) LSTMB(32, 32); else LSTMB(32, 16);
    } else {
        LSTMB(64, 16);
    }
#undef LSTMB
    return dgates;
}
