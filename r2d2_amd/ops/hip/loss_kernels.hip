// Fused double-Q target / TD / loss / priority kernels (gfx950).
//
// Replaces the reference learner's chain of eager ops + CPU numpy
// (argmax+gather worker.py:346-352, rescaled target :349, MSE/Huber loss
// :354, |TD| -> CPU :357, mixed priorities :359 + worker.py:268-276) with
// two kernels that keep everything on-device and also emit dLoss/dQ for the
// backward pass in the same sweep.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// ---------------------------------------------------------------------------
// Kernel 1: per learning-position row
//   a* = argmax_a q_online_tgt[row]          (first-max tie-break like torch)
//   qn = q_target_tgt[row][a*]
//   target = h(r + gamma_n * h^-1(qn))
//   td = q_learn[row][action] - target
//   loss_row = w * L(td);  dq[row][action] = w * L'(td) / R;  abs_td = |td|
// One thread per row (A <= 32); block-level reduction of the loss sum.
// ---------------------------------------------------------------------------
template <int LOSS_KIND>  // 0 = mse, 1 = huber
__global__ void fused_double_q_loss_kernel(
    const float* __restrict__ q_learn,       // (R, A)
    const float* __restrict__ q_online_tgt,  // (R, A)
    const float* __restrict__ q_target_tgt,  // (R, A)
    const long* __restrict__ action,         // (R,)
    const float* __restrict__ n_step_reward, // (R,)
    const float* __restrict__ gamma_n,       // (R,)
    const float* __restrict__ is_weights,    // (R,)
    float* __restrict__ dq,                  // (R, A) out (zero-filled here)
    float* __restrict__ abs_td,              // (R,) out
    float* __restrict__ target_out,          // (R,) out (for tests/priority)
    float* __restrict__ loss_sum,            // (1,) out, pre-zeroed
    int R, int A, float eps, float kappa) {
    int row = blockIdx.x * blockDim.x + threadIdx.x;
    float my_loss = 0.f;
    if (row < R) {
        const float* qo = q_online_tgt + (long)row * A;
        const float* qt = q_target_tgt + (long)row * A;
        // first-max argmax over the online net's target-position Q
        int a_star = 0;
        float best = qo[0];
        for (int a = 1; a < A; ++a) {
            float v = qo[a];
            if (v > best) { best = v; a_star = a; }
        }
        float qn = qt[a_star];
        float target = value_rescale(
            n_step_reward[row] + gamma_n[row] * inv_value_rescale(qn, eps), eps);

        long act = action[row];
        float q = q_learn[(long)row * A + act];
        float td = q - target;
        float w = is_weights[row];

        float l, dl;
        if (LOSS_KIND == 0) {            // mse (reference-compat)
            l = td * td;
            dl = 2.f * td;
        } else {                          // huber
            float at = fabsf(td);
            if (at <= kappa) { l = 0.5f * td * td; dl = td; }
            else { l = kappa * (at - 0.5f * kappa); dl = (td > 0.f ? kappa : -kappa); }
        }
        my_loss = w * l;

        float* dqr = dq + (long)row * A;
        for (int a = 0; a < A; ++a) dqr[a] = 0.f;
        dqr[act] = w * dl / (float)R;     // d(mean_i w_i L_i)/dq

        abs_td[row] = fabsf(td);
        target_out[row] = target;
    }
    // block reduce the loss sum, one atomic per block
    __shared__ float warp_sums[16];
    float ws = wave_reduce_sum(my_loss);
    int lane = threadIdx.x & (WAVE - 1);
    int wid = threadIdx.x / WAVE;
    if (lane == 0) warp_sums[wid] = ws;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int i = 0; i < (int)(blockDim.x / WAVE); ++i) s += warp_sums[i];
        atomicAdd(loss_sum, s);
    }
}

// ---------------------------------------------------------------------------
// Kernel 2: per-sequence mixed priority over ragged segments
//   prio[b] = eta * max(|td| seg) + (1-eta) * mean(|td| seg)
// One wave per segment; lanes stride the segment.
// ---------------------------------------------------------------------------
__global__ void segment_priority_kernel(
    const float* __restrict__ abs_td,      // (R,)
    const int* __restrict__ seg_offsets,   // (B+1,)
    float* __restrict__ prio,              // (B,) out
    int B, float eta) {
    int seg = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    if (seg >= B) return;
    int s = seg_offsets[seg], e = seg_offsets[seg + 1];
    float m = -INFINITY, acc = 0.f;
    for (int i = s + lane; i < e; i += WAVE) {
        float v = abs_td[i];
        m = fmaxf(m, v);
        acc += v;
    }
    m = wave_allreduce_max(m);
    acc = wave_allreduce_sum(acc);
    if (lane == 0) {
        int n = e - s;
        prio[seg] = (n > 0) ? (eta * m + (1.f - eta) * acc / (float)n) : 0.f;
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> fused_double_q_loss(
    torch::Tensor q_learn, torch::Tensor q_online_tgt, torch::Tensor q_target_tgt,
    torch::Tensor action, torch::Tensor n_step_reward, torch::Tensor gamma_n,
    torch::Tensor is_weights, double eps, double kappa, int64_t loss_kind) {
    TORCH_CHECK(q_learn.is_cuda() && q_learn.dtype() == torch::kFloat32);
    TORCH_CHECK(q_learn.is_contiguous() && q_online_tgt.is_contiguous()
                && q_target_tgt.is_contiguous());
    int R = q_learn.size(0), A = q_learn.size(1);
    TORCH_CHECK(A <= 32, "A must be <= 32");
    auto opts = q_learn.options();
    auto dq = torch::empty({R, A}, opts);
    auto abs_td = torch::empty({R}, opts);
    auto target = torch::empty({R}, opts);
    auto loss_sum = torch::zeros({1}, opts);

    const int threads = 256;
    int blocks = (R + threads - 1) / threads;
    auto stream = at::cuda::getCurrentCUDAStream();
    auto launch = [&](auto kernel) {
        hipLaunchKernelGGL(kernel, dim3(blocks), dim3(threads), 0, stream.stream(),
            q_learn.data_ptr<float>(), q_online_tgt.data_ptr<float>(),
            q_target_tgt.data_ptr<float>(), action.data_ptr<long>(),
            n_step_reward.data_ptr<float>(), gamma_n.data_ptr<float>(),
            is_weights.data_ptr<float>(), dq.data_ptr<float>(),
            abs_td.data_ptr<float>(), target.data_ptr<float>(),
            loss_sum.data_ptr<float>(), R, A, (float)eps, (float)kappa);
    };
    if (loss_kind == 0) launch(fused_double_q_loss_kernel<0>);
    else launch(fused_double_q_loss_kernel<1>);
    auto loss = loss_sum / R;  // mean
    return {loss, dq, abs_td, target};
}

torch::Tensor segment_priority(torch::Tensor abs_td, torch::Tensor seg_offsets,
                               double eta) {
    TORCH_CHECK(abs_td.is_cuda() && abs_td.dtype() == torch::kFloat32);
    TORCH_CHECK(seg_offsets.dtype() == torch::kInt32 && seg_offsets.is_cuda());
    int B = seg_offsets.size(0) - 1;
    auto prio = torch::empty({B}, abs_td.options());
    const int waves_per_block = 4;
    const int threads = WAVE * waves_per_block;
    int blocks = (B + waves_per_block - 1) / waves_per_block;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(segment_priority_kernel, dim3(blocks), dim3(threads), 0,
                       stream.stream(), abs_td.data_ptr<float>(),
                       seg_offsets.data_ptr<int>(), prio.data_ptr<float>(),
                       B, (float)eta);
    return prio;
}

// ---------------------------------------------------------------------------
// Dueling head combine: q[r][a] = v[r] + adv[r][a] - mean_{a<A}(adv[r][:A])
// adv/val come from padded 32-col head GEMMs (cols >= A are zero-weighted).
// Backward: dadv[a] = dq[a] - sum(dq)/A (a < A, else 0); dval = sum(dq).
// ---------------------------------------------------------------------------
__global__ void dueling_combine_kernel(
    const float* __restrict__ adv,   // (R, PADA)
    const float* __restrict__ val,   // (R, PADV) — col 0 is V
    float* __restrict__ q,           // (R, A)
    int R, int A, int PADA, int PADV) {
    int r = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    if (r >= R) return;
    float a = (lane < A) ? adv[(long)r * PADA + lane] : 0.f;
    float s = wave_allreduce_sum(a) / A;
    if (lane < A) q[(long)r * A + lane] = val[(long)r * PADV] + a - s;
}

__global__ void dueling_combine_bwd_kernel(
    const float* __restrict__ dq,    // (R, A)
    __hip_bfloat16* __restrict__ dadv,  // (R, PADA) bf16 (feeds gemm bwd)
    __hip_bfloat16* __restrict__ dval,  // (R, PADV)
    int R, int A, int PADA, int PADV) {
    int r = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    if (r >= R) return;
    float d = (lane < A) ? dq[(long)r * A + lane] : 0.f;
    float s = wave_allreduce_sum(d);
    if (lane < PADA)
        dadv[(long)r * PADA + lane] =
            f2bf((lane < A) ? (d - s / A) : 0.f);
    if (lane < PADV)
        dval[(long)r * PADV + lane] = f2bf((lane == 0) ? s : 0.f);
}

// ---------------------------------------------------------------------------
// scatter_dh: build the LSTM's upstream gradient dHext (BT, H) from the
// head-backward rows in ONE pass:
//   dHext[bt] = dh_a[row_of[bt]] + dh_v[row_of[bt]]   (row_of[bt] >= 0)
//             = 0                                      (no learning row at bt)
// Learn positions are unique per (b, t), so no atomics; replaces a zeros
// fill + two index_add_ launches + two bf16->f32 casts.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void scatter_dh_kernel(
    const __hip_bfloat16* __restrict__ dh_a,  // (R, H)
    const __hip_bfloat16* __restrict__ dh_v,  // (R, H)
    const int* __restrict__ row_of,           // (BT,) learn row or -1
    float* __restrict__ out,                  // (BT, H)
    long total, int H) {
    long idx = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    if (idx >= total) return;
    int bt = (int)(idx / H);
    int k = (int)(idx % H);
    int r = row_of[bt];
    float4 v{0.f, 0.f, 0.f, 0.f};
    if (r >= 0) {
        long off = (long)r * H + k;
        v.x = bf2f(dh_a[off]) + bf2f(dh_v[off]);
        v.y = bf2f(dh_a[off + 1]) + bf2f(dh_v[off + 1]);
        v.z = bf2f(dh_a[off + 2]) + bf2f(dh_v[off + 2]);
        v.w = bf2f(dh_a[off + 3]) + bf2f(dh_v[off + 3]);
    }
    *reinterpret_cast<float4*>(out + idx) = v;
}

torch::Tensor scatter_dh(torch::Tensor dh_a, torch::Tensor dh_v,
                         torch::Tensor row_of, int64_t BT) {
    TORCH_CHECK(dh_a.is_cuda() && dh_a.dtype() == torch::kBFloat16
                && dh_a.is_contiguous() && dh_v.is_contiguous());
    TORCH_CHECK(row_of.dtype() == torch::kInt32 && row_of.numel() == BT);
    long H = dh_a.size(1);
    TORCH_CHECK(H % 4 == 0);
    auto out = torch::empty({BT, H},
                            dh_a.options().dtype(torch::kFloat32));
    long total = BT * H;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(scatter_dh_kernel,
                       dim3((int)((total / 4 + 255) / 256)), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(dh_a.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(dh_v.data_ptr()),
                       row_of.data_ptr<int>(), out.data_ptr<float>(),
                       total, (int)H);
    return out;
}

torch::Tensor dueling_combine(torch::Tensor adv, torch::Tensor val, int64_t A) {
    int R = adv.size(0), PADA = adv.size(1), PADV = val.size(1);
    auto q = torch::empty({R, A}, adv.options());
    const int wpb = 4;
    dim3 grid((R + wpb - 1) / wpb);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dueling_combine_kernel, grid, dim3(WAVE * wpb), 0,
                       stream.stream(), adv.data_ptr<float>(),
                       val.data_ptr<float>(), q.data_ptr<float>(), R, (int)A,
                       PADA, PADV);
    return q;
}

std::vector<torch::Tensor> dueling_combine_bwd(torch::Tensor dq, int64_t PADA,
                                               int64_t PADV) {
    int R = dq.size(0), A = dq.size(1);
    auto opts = dq.options().dtype(torch::kBFloat16);
    auto dadv = torch::empty({R, PADA}, opts);
    auto dval = torch::empty({R, PADV}, opts);
    const int wpb = 4;
    dim3 grid((R + wpb - 1) / wpb);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dueling_combine_bwd_kernel, grid, dim3(WAVE * wpb), 0,
                       stream.stream(), dq.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dadv.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(dval.data_ptr()),
                       R, A, (int)PADA, (int)PADV);
    return {dadv, dval};
}
