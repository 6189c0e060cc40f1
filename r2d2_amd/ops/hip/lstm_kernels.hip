// Fused persistent LSTM kernels (gfx950) — SURVEY.md §2.3 K6, the hard one.
//
// Replaces nn.LSTM + pack_padded_sequence (reference model.py:95-100,136-141;
// 87k hipBLASLt launches per profile) with ONE kernel per direction:
//
// - The input GEMM X = rin @ W_ih^T + b is precomputed for all (B, T) steps
//   by the MFMA GEMM (gemm_kernels.hip) — it has no sequential dependence.
// - The recurrent part runs as a single persistent launch: each workgroup
//   owns a slice of hidden units, keeps its W_hh slice resident in LDS for
//   all T steps, computes its gate columns with MFMA from the previous
//   step's h (read through L2), applies the gate nonlinearities, and
//   advances h/c.  Steps are separated by an XCD-sharded global barrier
//   (agent-scope release/acquire, relaxed polling + s_sleep, cumulative
//   epoch counters zeroed by hipMemsetAsync before every launch — HIP guide
//   §6 G16).  Per-sample length masks replace pack_padded semantics: masked
//   steps copy h/c through unchanged.
// - Two networks (online + target) ride in one launch (disjoint workgroup
//   ranges), sharing the per-step barrier.
// - Backward (online only) runs the reverse-time recurrence: per step one
//   MFMA GEMM dh += dgates_{t+1} @ W_hh with the wg's W_hh^T slice LDS-
//   resident, elementwise gate backward, dgates written to a global stash.
//   The big weight gradients (dW_hh, dW_ih, db) and dX are then plain
//   GEMMs outside the kernel (gemm_wgrad / gemm_dgrad over B*T rows).
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
    __bf16 e[8];
};

__device__ __forceinline__ bf16x8 lload8(const __hip_bfloat16* p) {
    lbf8u r;
    r.u = *reinterpret_cast<const uint4*>(p);
    return r.v;
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
// XCD-sharded grid barrier (plain launch; grid <= 256 blocks => resident).
// Cumulative counters: epoch e complete when bucket counts reach e*per_bucket
// and top count reaches e*8.  One release fence before arrival, one acquire
// after the generation flip; relaxed polling with s_sleep.
// ---------------------------------------------------------------------------
struct GridBar {
    unsigned bucket[8];
    unsigned top;
    unsigned gen;
    unsigned poison;
};

__device__ __forceinline__ bool grid_barrier(GridBar* bar, unsigned epoch,
                                             int nblocks) {
    __syncthreads();
    __shared__ unsigned ok_s;
    if (threadIdx.x == 0) {
        int b = blockIdx.x & 7;
        int per = (nblocks + 7 - b) >> 3;  // blocks with id%8 == b
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
            __builtin_amdgcn_s_sleep(8);
            if (++spins > (long)2e8) {  // bounded spin: poison and bail
                __hip_atomic_store(&bar->poison, 1u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
                ok = 0;
                break;
            }
            if (__hip_atomic_load(&bar->poison, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT)) {
                ok = 0;
                break;
            }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        ok_s = ok;
    }
    __syncthreads();
    return ok_s != 0;
}

// ---------------------------------------------------------------------------
// Forward.  Grid: wgs_per_net = H/8 blocks per network (net1 optional).
// Each wg: 8 hidden units -> 32 gate columns, W_hh slice (32 x H) in LDS.
// Per step: gates(B,32) = h_prev(B,H) @ Whh_slice^T + X[t] -> c,h update.
// ---------------------------------------------------------------------------
template <int H>
__global__ __launch_bounds__(256) void lstm_fwd_kernel(
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
    GridBar* bar, int B, int T, int nblocks) {
    constexpr int WGS_PER_NET = H / 8;
    const int net = blockIdx.x / WGS_PER_NET;
    const int wid = blockIdx.x % WGS_PER_NET;
    const int u0 = wid * 8;

    const __hip_bfloat16* X = net ? X1 : X0;
    const __hip_bfloat16* Whh = net ? Whh1 : Whh0;
    const float* init = net ? init1 : init0;
    __hip_bfloat16* Hout = net ? Hout1 : Hout0;
    float* Cout = net ? Cout1 : Cout0;
    __hip_bfloat16* stash = net ? nullptr : stash0;

    __shared__ __hip_bfloat16 s_whh[32][H + 8];
    __shared__ float s_gates[64][32 + 4];

    // stage the wg's 32 W_hh rows (gate g, unit u0+j -> row g*H + u0 + j;
    // local col c = g*8 + j)
    for (int e = threadIdx.x * 8; e < 32 * H; e += blockDim.x * 8) {
        int c = e / H;
        int k = e % H;
        int g = c / 8, j = c % 8;
        *reinterpret_cast<bf16x8*>(&s_whh[c][k]) =
            lload8(Whh + (long)(g * H + u0 + j) * H + k);
    }
    // write h0/c0 into the output buffers (the wg's 8 units)
    for (int p = threadIdx.x; p < B * 8; p += blockDim.x) {
        int b = p / 8, j = p % 8;
        int u = u0 + j;
        Hout[((long)b * (T + 1)) * H + u] = f2bf(init[(long)b * H + u]);
        Cout[((long)b * (T + 1)) * H + u] = init[((long)B + b) * H + u];
    }
    if (!grid_barrier(bar, 1, nblocks)) return;

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wrow0 = (wave >> 1) * 32;        // rows 0..63 (batch)
    const int wcol0 = (wave & 1) * 16;         // cols 0..31 (gate local)
    const int frow = lane & 15;
    const int kseg = (lane >> 4) * 8;

    for (int t = 0; t < T; ++t) {
        // gates = h_prev @ Whh_slice^T
        f32x4 acc[2] = {};
#pragma unroll 4
        for (int k0 = 0; k0 < H; k0 += 32) {
            bf16x8 bfr = lload8(&s_whh[wcol0 + frow][k0 + kseg]);
#pragma unroll
            for (int i = 0; i < 2; ++i) {
                int row = wrow0 + i * 16 + frow;
                bf16x8 afr = (row < B)
                    ? lload8(Hout + ((long)row * (T + 1) + t) * H + k0 + kseg)
                    : lzero8();
                acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr, bfr, acc[i], 0, 0, 0);
            }
        }
        // + X[t], stage to LDS
        {
            int ccol = lane & 15;
            int crow = (lane >> 4) * 4;
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int row = wrow0 + i * 16 + crow + r;
                    int c = wcol0 + ccol;
                    if (row < B) {
                        int g = c / 8, j = c % 8;
                        float x = bf2f(X[((long)row * T + t) * 4 * H
                                         + g * H + u0 + j]);
                        s_gates[row][c] = acc[i][r] + x;
                    }
                }
        }
        __syncthreads();
        // gate nonlinearities + state advance for (b, j) pairs
        for (int p = threadIdx.x; p < B * 8; p += blockDim.x) {
            int b = p / 8, j = p % 8;
            int u = u0 + j;
            long prev_off = ((long)b * (T + 1) + t) * H + u;
            long cur_off = prev_off + H;
            float c_prev = Cout[prev_off];
            bool active = t < lens[b];
            float i_ = 0.f, f_ = 0.f, g_ = 0.f, o_ = 0.f, c, h;
            if (active) {
                i_ = sigmoidf_(s_gates[b][0 + j]);
                f_ = sigmoidf_(s_gates[b][8 + j]);
                g_ = tanhf(s_gates[b][16 + j]);
                o_ = sigmoidf_(s_gates[b][24 + j]);
                c = f_ * c_prev + i_ * g_;
                h = o_ * tanhf(c);
            } else {
                c = c_prev;
                h = bf2f(Hout[prev_off]);
            }
            Cout[cur_off] = c;
            Hout[cur_off] = f2bf(h);
            if (stash) {
                long so = ((long)b * T + t) * 4 * H + u;
                stash[so] = f2bf(i_);
                stash[so + H] = f2bf(f_);
                stash[so + 2 * H] = f2bf(g_);
                stash[so + 3 * H] = f2bf(o_);
            }
        }
        if (!grid_barrier(bar, (unsigned)(t + 2), nblocks)) return;
    }
}

// ---------------------------------------------------------------------------
// Backward (online net).  Grid: H/16 wgs, each 16 hidden units; W_hh^T
// slice (16 x 4H) in LDS.  Reverse loop; dgates -> global stash (zeros at
// masked steps), dh/dc kept per-wg in LDS.
// ---------------------------------------------------------------------------
template <int H>
__global__ __launch_bounds__(256) void lstm_bwd_kernel(
    const __hip_bfloat16* __restrict__ stash,  // (B, T, 4H) i,f,g,o
    const float* __restrict__ Cout,            // (B, T+1, H)
    const __hip_bfloat16* __restrict__ Hout,   // (B, T+1, H) (unused, kept)
    const float* __restrict__ dHext,           // (B, T, H) upstream
    const __hip_bfloat16* __restrict__ Whh_bwd,// (H, 4H): W_hh^T row-major
    const int* __restrict__ lens,
    __hip_bfloat16* __restrict__ dgates,       // (B, T, 4H) out
    GridBar* bar, int B, int T, int nblocks) {
    const int wid = blockIdx.x;
    const int u0 = wid * 16;

    __shared__ __hip_bfloat16 s_wb[16][4 * H + 8];
    __shared__ float s_dh[64][16 + 1];
    __shared__ float s_dc[64][16 + 1];
    __shared__ float s_rec[64][16 + 1];

    for (int e = threadIdx.x * 8; e < 16 * 4 * H; e += blockDim.x * 8) {
        int c = e / (4 * H);
        int k = e % (4 * H);
        *reinterpret_cast<bf16x8*>(&s_wb[c][k]) =
            lload8(Whh_bwd + (long)(u0 + c) * 4 * H + k);
    }
    for (int p = threadIdx.x; p < B * 16; p += blockDim.x) {
        s_dh[p / 16][p % 16] = 0.f;
        s_dc[p / 16][p % 16] = 0.f;
    }
    __syncthreads();

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wrow0 = wave * 16;   // 4 waves x 16 rows = 64 batch rows
    const int frow = lane & 15;
    const int kseg = (lane >> 4) * 8;

    for (int t = T - 1; t >= 0; --t) {
        // recurrent contribution: rec(B,16) = dgates_{t+1}(B,4H) @ s_wb^T
        if (t < T - 1) {
            f32x4 acc = {};
            for (int k0 = 0; k0 < 4 * H; k0 += 32) {
                bf16x8 bfr = lload8(&s_wb[frow][k0 + kseg]);
                int row = wrow0 + frow;
                bf16x8 afr = (row < B)
                    ? lload8(dgates + ((long)row * T + t + 1) * 4 * H + k0 + kseg)
                    : lzero8();
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr, bfr, acc,
                                                              0, 0, 0);
            }
            int ccol = lane & 15;
            int crow = (lane >> 4) * 4;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = wrow0 + crow + r;
                if (row < B) s_rec[row][ccol] = acc[r];
            }
        }
        __syncthreads();

        for (int p = threadIdx.x; p < B * 16; p += blockDim.x) {
            int b = p / 16, jl = p % 16;
            int u = u0 + jl;
            float dh, dc_in;
            bool last = (t == T - 1);
            bool active_next = !last && ((t + 1) < lens[b]);
            float ext = dHext[((long)b * T + t) * H + u];
            if (last) {
                dh = ext;
                dc_in = 0.f;
            } else if (active_next) {
                long so1 = ((long)b * T + t + 1) * 4 * H + u;
                float f_next = bf2f(stash[so1 + H]);
                dh = ext + s_rec[b][jl];
                dc_in = s_dc[b][jl] * f_next;
            } else {
                dh = ext + s_dh[b][jl];
                dc_in = s_dc[b][jl];
            }
            bool active = t < lens[b];
            long so = ((long)b * T + t) * 4 * H + u;
            if (active) {
                float i_ = bf2f(stash[so]);
                float f_ = bf2f(stash[so + H]);
                float g_ = bf2f(stash[so + 2 * H]);
                float o_ = bf2f(stash[so + 3 * H]);
                float tc = tanhf(Cout[((long)b * (T + 1) + t + 1) * H + u]);
                float c_prev = Cout[((long)b * (T + 1) + t) * H + u];
                float dc = dc_in + dh * o_ * (1.f - tc * tc);
                dgates[so] = f2bf(dc * g_ * i_ * (1.f - i_));
                dgates[so + H] = f2bf(dc * c_prev * f_ * (1.f - f_));
                dgates[so + 2 * H] = f2bf(dc * i_ * (1.f - g_ * g_));
                dgates[so + 3 * H] = f2bf(dh * tc * o_ * (1.f - o_));
                s_dh[b][jl] = dh;
                s_dc[b][jl] = dc;
            } else {
                dgates[so] = (__hip_bfloat16)0.f;
                dgates[so + H] = (__hip_bfloat16)0.f;
                dgates[so + 2 * H] = (__hip_bfloat16)0.f;
                dgates[so + 3 * H] = (__hip_bfloat16)0.f;
                s_dh[b][jl] = dh;
                s_dc[b][jl] = dc_in;
            }
        }
        if (!grid_barrier(bar, (unsigned)(T - t), nblocks)) return;
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

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
    auto stash = want_stash ? torch::empty({B, T, H4}, bf)
                            : torch::Tensor();

    int wgs = (int)H / 8;
    int nblocks = wgs * (two ? 2 : 1);
    auto stream = at::cuda::getCurrentCUDAStream();
    TORCH_CHECK(barrier_ws.numel() * barrier_ws.element_size()
                >= (long)sizeof(GridBar));
    hipMemsetAsync(barrier_ws.data_ptr(), 0, sizeof(GridBar), stream.stream());

    auto bp = [](torch::Tensor& t) {
        return t.defined()
            ? reinterpret_cast<__hip_bfloat16*>(t.data_ptr()) : nullptr;
    };
    hipLaunchKernelGGL((lstm_fwd_kernel<512>), dim3(nblocks), dim3(256), 0,
        stream.stream(),
        reinterpret_cast<const __hip_bfloat16*>(X0.data_ptr()),
        two ? reinterpret_cast<const __hip_bfloat16*>(X1.data_ptr()) : nullptr,
        reinterpret_cast<const __hip_bfloat16*>(Whh0.data_ptr()),
        two ? reinterpret_cast<const __hip_bfloat16*>(Whh1.data_ptr()) : nullptr,
        init0.data_ptr<float>(),
        two ? init1.data_ptr<float>() : nullptr,
        lens.data_ptr<int>(), bp(H0), bp(H1),
        C0.data_ptr<float>(), two ? C1.data_ptr<float>() : nullptr,
        want_stash ? bp(stash) : nullptr,
        reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),
        (int)B, (int)T, nblocks);

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
    int nblocks = (int)H / 16;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(barrier_ws.data_ptr(), 0, sizeof(GridBar), stream.stream());
    hipLaunchKernelGGL((lstm_bwd_kernel<512>), dim3(nblocks), dim3(256), 0,
        stream.stream(),
        reinterpret_cast<const __hip_bfloat16*>(stash.data_ptr()),
        Cout.data_ptr<float>(),
        reinterpret_cast<const __hip_bfloat16*>(Hout.data_ptr()),
        dHext.data_ptr<float>(),
        reinterpret_cast<const __hip_bfloat16*>(Whh_bwd.data_ptr()),
        lens.data_ptr<int>(),
        reinterpret_cast<__hip_bfloat16*>(dgates.data_ptr()),
        reinterpret_cast<GridBar*>(barrier_ws.data_ptr()),
        (int)B, (int)T, nblocks);
    return dgates;
}
