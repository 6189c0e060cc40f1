// GPU-resident prioritized-replay sum-tree (gfx950).
//
// Replaces the reference's CPU/numpy PriorityTree (priority_tree.py) with a
// device-resident heap-layout tree (f64 nodes, like the reference's float64)
// supporting concurrent-safe updates via atomic leaf-exchange + atomic delta
// propagation, stratified n-way sampling with batch-min IS weights in one
// kernel, and the ring-overwrite stale-priority mask of worker.py:247-256.
//
// Layout: one flat double array, node 0 = root, children of i = 2i+1/2i+2,
// leaves at offset 2^L - 1.  All kernels are tiny (M, n <= 1024); the tree
// lives in HBM and is hot in L2.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

__device__ __forceinline__ double atomic_exch_f64(double* addr, double val) {
    unsigned long long old = atomicExch(
        reinterpret_cast<unsigned long long*>(addr),
        __double_as_longlong(val));
    return __longlong_as_double(old);
}

// ---------------------------------------------------------------------------
// update: leaves[idx] = td^alpha, parents += delta (atomic, duplicate-safe).
// The stale mask skips indexes whose ring block was overwritten between
// sampling (old_ptr) and this update (cur_ptr); pass old_ptr == cur_ptr to
// disable (ingest path).
// ---------------------------------------------------------------------------
__global__ void sumtree_update_kernel(
    double* __restrict__ tree, long leaf_offset,
    const long* __restrict__ idxes, const float* __restrict__ td,
    int M, float alpha, long old_ptr, long cur_ptr, long seq_per_block,
    long num_blocks) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= M) return;
    long idx = idxes[i];
    if (old_ptr != cur_ptr) {
        long lo = old_ptr * seq_per_block, hi = cur_ptr * seq_per_block;
        bool stale = (cur_ptr > old_ptr) ? (idx >= lo && idx < hi)
                                         : (idx >= lo || idx < hi);
        if (stale) return;
    }
    double p = pow((double)fabsf(td[i]), (double)alpha);
    long node = leaf_offset + idx;
    double old = atomic_exch_f64(&tree[node], p);
    double delta = p - old;
    while (node > 0) {
        node = (node - 1) >> 1;
        atomicAdd(&tree[node], delta);
    }
}

// ---------------------------------------------------------------------------
// sample: stratified descent, one thread per draw, single block (n <= 1024);
// IS weights normalized by the batch min (reference priority_tree.py:39-41).
// ---------------------------------------------------------------------------
__global__ void sumtree_sample_kernel(
    const double* __restrict__ tree, long leaf_offset, int num_levels,
    const float* __restrict__ jitter,   // (n,) uniform[0,1)
    long* __restrict__ out_idx,         // (n,)
    float* __restrict__ out_prio,       // (n,)
    float* __restrict__ out_weight,     // (n,)
    int n, float beta, long max_idx) {
    int i = threadIdx.x;
    __shared__ double warp_min[16];
    double prio = 0.0;
    long idx = 0;
    if (i < n) {
        double total = tree[0];
        double u = (i + (double)jitter[i]) * (total / n);
        long node = 0;
        for (int l = 0; l < num_levels; ++l) {
            long left = 2 * node + 1;
            double lsum = tree[left];
            if (u < lsum) {
                node = left;
            } else {
                u -= lsum;
                node = left + 1;
            }
        }
        idx = node - leaf_offset;
        // fp-edge guard: a rounding overshoot in the subtract path could
        // land in the zero-padded leaves past the last real sequence
        if (idx > max_idx) idx = max_idx;
        prio = tree[leaf_offset + idx];
        out_idx[i] = idx;
        out_prio[i] = (float)prio;
    }
    // batch min of positive priorities (fp-edge zero leaves excluded)
    double m = (i < n && prio > 0.0) ? prio : INFINITY;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmin(m, __shfl_xor(m, off));
    int lane = threadIdx.x & (WAVE - 1);
    int wid = threadIdx.x / WAVE;
    if (lane == 0) warp_min[wid] = m;
    __syncthreads();
    if (threadIdx.x == 0) {
        double mm = INFINITY;
        for (int w = 0; w < (int)(blockDim.x + WAVE - 1) / WAVE; ++w)
            mm = fmin(mm, warp_min[w]);
        warp_min[0] = (mm == INFINITY) ? 1.0 : mm;
    }
    __syncthreads();
    if (i < n) {
        double minp = warp_min[0];
        double p = fmax(prio, minp * 1e-12);
        out_weight[i] = (float)pow(p / minp, (double)-beta);
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

void sumtree_update(torch::Tensor tree, int64_t leaf_offset, torch::Tensor idxes,
                    torch::Tensor td, double alpha, int64_t old_ptr,
                    int64_t cur_ptr, int64_t seq_per_block, int64_t num_blocks) {
    TORCH_CHECK(tree.is_cuda() && tree.dtype() == torch::kFloat64);
    TORCH_CHECK(idxes.dtype() == torch::kInt64 && td.dtype() == torch::kFloat32);
    int M = idxes.size(0);
    if (M == 0) return;
    const int threads = 256;
    int blocks = (M + threads - 1) / threads;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(sumtree_update_kernel, dim3(blocks), dim3(threads), 0,
                       stream.stream(), tree.data_ptr<double>(), leaf_offset,
                       idxes.data_ptr<long>(), td.data_ptr<float>(), M,
                       (float)alpha, old_ptr, cur_ptr, seq_per_block, num_blocks);
}

std::vector<torch::Tensor> sumtree_sample(torch::Tensor tree, int64_t leaf_offset,
                                          int64_t num_levels, torch::Tensor jitter,
                                          int64_t n, double beta,
                                          int64_t max_idx) {
    TORCH_CHECK(tree.is_cuda() && tree.dtype() == torch::kFloat64);
    TORCH_CHECK(n <= 1024, "sample batch must be <= 1024");
    auto opts_l = tree.options().dtype(torch::kInt64);
    auto opts_f = tree.options().dtype(torch::kFloat32);
    auto idx = torch::empty({n}, opts_l);
    auto prio = torch::empty({n}, opts_f);
    auto weight = torch::empty({n}, opts_f);
    int threads = 64;
    while (threads < n) threads *= 2;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(sumtree_sample_kernel, dim3(1), dim3(threads), 0,
                       stream.stream(), tree.data_ptr<double>(), leaf_offset,
                       (int)num_levels, jitter.data_ptr<float>(),
                       idx.data_ptr<long>(), prio.data_ptr<float>(),
                       weight.data_ptr<float>(), (int)n, (float)beta,
                       (long)max_idx);
    return {idx, prio, weight};
}
