// Fused optimizer kernels (gfx950) — SURVEY.md §2.3 K12/K13.
//
// The engine keeps ALL parameters in one flat f32 buffer (module params are
// views), gradients in a matching flat buffer, so grad-norm clip + Adam for
// the whole 4.3M-param network is TWO kernels (the reference runs
// clip_grad_norm_ + Adam over ~20 tensors, worker.py:364-365):
//   1. grad_sumsq: grid-stride f32x4 squared-sum reduction -> norm_buf[0]
//   2. adam_step: elementwise Adam with the clip scale
//      min(1, max_norm/norm) computed in-kernel from norm_buf, matching
//      torch.nn.utils.clip_grad_norm_ + torch.optim.Adam numerics.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4v;

__global__ void grad_sumsq_kernel(const float* __restrict__ g, long n,
                                  float* __restrict__ out) {
    long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    long stride = (long)gridDim.x * blockDim.x * 4;
    float acc = 0.f;
    for (long i = i0; i + 4 <= n; i += stride) {
        f32x4v v = *reinterpret_cast<const f32x4v*>(g + i);
        acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    // tail
    if (i0 == 0) {
        for (long i = (n / 4) * 4; i < n; ++i) acc += g[i] * g[i];
    }
    acc = wave_reduce_sum(acc);
    __shared__ float ws[16];
    int lane = threadIdx.x & (WAVE - 1);
    int wid = threadIdx.x / WAVE;
    if (lane == 0) ws[wid] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int i = 0; i < (int)(blockDim.x / WAVE); ++i) s += ws[i];
        atomicAdd(out, s);
    }
}

__global__ void adam_step_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ norm_buf, long n,
    float max_norm, float lr, float beta1, float beta2, float eps,
    float bias_c1, float bias_c2) {
    float scale = 1.f;
    if (max_norm > 0.f) {
        float norm = sqrtf(norm_buf[0]);
        // torch clip_grad_norm_: scale = max_norm / (norm + 1e-6), capped at 1
        float sc = max_norm / (norm + 1e-6f);
        scale = fminf(1.f, sc);
    }
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float inv_c1 = 1.f / bias_c1;
    float inv_sqrt_c2 = 1.f / sqrtf(bias_c2);
    for (; i < n; i += stride) {
        float gi = g[i] * scale;
        float mi = beta1 * m[i] + (1.f - beta1) * gi;
        float vi = beta2 * v[i] + (1.f - beta2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        float denom = sqrtf(vi) * inv_sqrt_c2 + eps;
        p[i] -= lr * inv_c1 * mi / denom;
    }
}

// ---------------------------------------------------------------------------
// gather_pack: regenerate ALL prepacked weight tensors from the flat f32
// parameter buffer in ONE launch per output dtype.  Each packed element is
// a permutation (map1) of flat_param, optionally plus a second source
// (map2 — the LSTM bias sum b_ih + b_hh); -1 in map1 means structural zero
// padding.  Replaces the ~40 permute/cast/zero launches of the python
// repack that otherwise run EVERY optimizer step (engine refresh_online).
// ---------------------------------------------------------------------------
template <bool OUT_BF16>
__global__ void gather_pack_kernel(
    const float* __restrict__ flat, const int* __restrict__ m1,
    const int* __restrict__ m2, void* __restrict__ out, long n) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int a = m1[i];
        float v = (a >= 0) ? flat[a] : 0.f;
        int b = m2[i];
        if (b >= 0) v += flat[b];
        if (OUT_BF16)
            reinterpret_cast<__hip_bfloat16*>(out)[i] = f2bf(v);
        else
            reinterpret_cast<float*>(out)[i] = v;
    }
}

void gather_pack(torch::Tensor flat, torch::Tensor m1, torch::Tensor m2,
                 torch::Tensor out) {
    TORCH_CHECK(flat.is_cuda() && flat.dtype() == torch::kFloat32);
    TORCH_CHECK(m1.dtype() == torch::kInt32 && m2.dtype() == torch::kInt32);
    long n = out.numel();
    if (n == 0) return;
    TORCH_CHECK(m1.numel() == n && m2.numel() == n);
    auto stream = at::cuda::getCurrentCUDAStream();
    int blocks = (int)std::min(2048L, (n + 255) / 256);
    if (out.dtype() == torch::kBFloat16)
        hipLaunchKernelGGL(gather_pack_kernel<true>, dim3(blocks), dim3(256),
                           0, stream.stream(), flat.data_ptr<float>(),
                           m1.data_ptr<int>(), m2.data_ptr<int>(),
                           out.data_ptr(), n);
    else
        hipLaunchKernelGGL(gather_pack_kernel<false>, dim3(blocks), dim3(256),
                           0, stream.stream(), flat.data_ptr<float>(),
                           m1.data_ptr<int>(), m2.data_ptr<int>(),
                           out.data_ptr(), n);
}

torch::Tensor grad_sumsq(torch::Tensor grad, torch::Tensor norm_buf) {
    TORCH_CHECK(grad.is_cuda() && grad.dtype() == torch::kFloat32
                && grad.is_contiguous());
    long n = grad.numel();
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(norm_buf.data_ptr(), 0, 4, stream.stream());
    int blocks = (int)std::min(2048L, (n / 4 + 255) / 256);
    hipLaunchKernelGGL(grad_sumsq_kernel, dim3(std::max(1, blocks)), dim3(256),
                       0, stream.stream(), grad.data_ptr<float>(), n,
                       norm_buf.data_ptr<float>());
    return norm_buf;
}

void adam_step(torch::Tensor param, torch::Tensor grad, torch::Tensor m,
               torch::Tensor v, torch::Tensor norm_buf, double max_norm,
               double lr, double beta1, double beta2, double eps,
               int64_t step) {
    long n = param.numel();
    double bc1 = 1.0 - std::pow(beta1, (double)step);
    double bc2 = 1.0 - std::pow(beta2, (double)step);
    auto stream = at::cuda::getCurrentCUDAStream();
    int blocks = (int)std::min(2048L, (n + 255) / 256);
    hipLaunchKernelGGL(adam_step_kernel, dim3(blocks), dim3(256), 0,
                       stream.stream(), param.data_ptr<float>(),
                       grad.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), norm_buf.data_ptr<float>(), n,
                       (float)max_norm, (float)lr, (float)beta1, (float)beta2,
                       (float)eps, (float)bc1, (float)bc2);
}
