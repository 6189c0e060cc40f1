// On-device batch assembly from the GPU-resident block store (gfx950).
//
// Replaces the reference's single-threaded Python slice loop + pad_sequence
// (worker.py:176-214) with device gather kernels reading directly from the
// HBM block store the sampled sequences live in.  No host round-trip: the
// sum-tree sample output feeds these kernels on the same stream.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// ---------------------------------------------------------------------------
// Kernel A: per-sample sequence metadata + segment offsets (single block).
//   idx -> (block, seq); read per-seq meta; exclusive-scan learn -> seg.
// ---------------------------------------------------------------------------
__global__ void gather_meta_kernel(
    const long* __restrict__ idx,          // (B,) sampled sequence indexes
    const int* __restrict__ burn_s,        // (num_blocks*spb,)
    const int* __restrict__ learn_s,
    const int* __restrict__ fwd_s,
    const int* __restrict__ obs_start_s,
    const int* __restrict__ learn_off_s,
    int* __restrict__ out_meta,            // (5, B): burn, learn, fwd, obs_start, learn_off
    int* __restrict__ seg,                 // (B+1,)
    int B, int spb) {
    __shared__ int scan_buf[1024];
    int i = threadIdx.x;
    int learn = 0;
    if (i < B) {
        long s = idx[i];
        out_meta[0 * B + i] = burn_s[s];
        learn = learn_s[s];
        out_meta[1 * B + i] = learn;
        out_meta[2 * B + i] = fwd_s[s];
        out_meta[3 * B + i] = obs_start_s[s];
        out_meta[4 * B + i] = learn_off_s[s];
    }
    scan_buf[i] = learn;
    __syncthreads();
    // inclusive scan (Hillis-Steele) over blockDim.x
    for (int off = 1; off < blockDim.x; off <<= 1) {
        int v = (i >= off) ? scan_buf[i - off] : 0;
        __syncthreads();
        scan_buf[i] += v;
        __syncthreads();
    }
    if (i < B) seg[i + 1] = scan_buf[i];
    if (i == 0) seg[0] = 0;
}

// ---------------------------------------------------------------------------
// Kernel B: padded sequence tensors.
//   grid.x = B * T; each block copies one (sample, time) frame (16B chunks)
//   and lane 0..A-1 fill the one-hot last_action, lane 0 the last_reward.
// ---------------------------------------------------------------------------
__global__ void gather_frames_kernel(
    const unsigned char* __restrict__ obs_store,  // (num_blocks, obs_rows, FB)
    const unsigned char* __restrict__ la_store,   // (num_blocks, obs_rows)
    const float* __restrict__ lr_store,           // (num_blocks, obs_rows)
    const int* __restrict__ meta,                 // (5, B)
    unsigned char* __restrict__ obs_out,          // (B, T, FB)
    float* __restrict__ la_out,                   // (B, T, A)
    float* __restrict__ lr_out,                   // (B, T)
    int B, int T, int A, long frame_bytes, long obs_rows, int spb) {
    int bt = blockIdx.x;
    int b = bt / T, t = bt % T;
    int burn = meta[0 * B + b], learn = meta[1 * B + b], fwd = meta[2 * B + b];
    int obs_start = meta[3 * B + b];
    int valid = burn + learn + fwd;
    long blk = 0;  // encoded inside obs_start: obs_start = block*obs_rows + row
    long row = (long)obs_start - burn + t;
    unsigned char* dst = obs_out + ((long)b * T + t) * frame_bytes;
    if (t < valid) {
        const unsigned char* src = obs_store + row * frame_bytes;
        for (long o = threadIdx.x * 16; o + 16 <= frame_bytes; o += blockDim.x * 16)
            *reinterpret_cast<uint4*>(dst + o) =
                *reinterpret_cast<const uint4*>(src + o);
        // frame_bytes tail (not multiple of 16)
        long tail = (frame_bytes / 16) * 16;
        for (long o = tail + threadIdx.x; o < frame_bytes; o += blockDim.x)
            dst[o] = src[o];
        if (threadIdx.x < A)
            la_out[((long)b * T + t) * A + threadIdx.x] =
                (la_store[row] == threadIdx.x) ? 1.f : 0.f;
        if (threadIdx.x == 0) lr_out[(long)b * T + t] = lr_store[row];
    } else {
        for (long o = threadIdx.x * 16; o + 16 <= frame_bytes; o += blockDim.x * 16)
            *reinterpret_cast<uint4*>(dst + o) = uint4{0, 0, 0, 0};
        long tail = (frame_bytes / 16) * 16;
        for (long o = tail + threadIdx.x; o < frame_bytes; o += blockDim.x)
            dst[o] = 0;
        if (threadIdx.x < A) la_out[((long)b * T + t) * A + threadIdx.x] = 0.f;
        if (threadIdx.x == 0) lr_out[(long)b * T + t] = 0.f;
    }
}

// ---------------------------------------------------------------------------
// Kernel C: flat per-learning-step arrays + hidden states + repeated weights.
//   grid over B * max_learn threads.
// ---------------------------------------------------------------------------
__global__ void gather_flat_kernel(
    const unsigned char* __restrict__ act_store,  // (num_blocks, block_len)
    const float* __restrict__ nsr_store,
    const float* __restrict__ gam_store,
    const float* __restrict__ hid_store,          // (num_blocks*spb, 2*H)
    const long* __restrict__ idx,                 // (B,)
    const int* __restrict__ meta,                 // (5, B)
    const int* __restrict__ seg,                  // (B+1,)
    const float* __restrict__ weights,            // (B,)
    long* __restrict__ act_out,                   // (Rmax,)
    float* __restrict__ nsr_out, float* __restrict__ gam_out,
    float* __restrict__ w_out,
    float* __restrict__ hid_out,                  // (2, B, H)
    int B, int max_learn, int H, long block_len, int spb) {
    int tid = blockIdx.x * blockDim.x + threadIdx.x;
    int b = tid / max_learn, j = tid % max_learn;
    if (b >= B) return;
    int learn = meta[1 * B + b];
    long s = idx[b];
    long blk = s / spb;
    if (j < learn) {
        long src = blk * block_len + meta[4 * B + b] + j;
        long dst = seg[b] + j;
        act_out[dst] = (long)act_store[src];
        nsr_out[dst] = nsr_store[src];
        gam_out[dst] = gam_store[src];
        w_out[dst] = weights[b];
    }
    // hidden: reuse threads j < 2*H strided
    for (int h = j; h < 2 * H; h += max_learn) {
        int which = h / H, hi = h % H;
        hid_out[((long)which * B + b) * H + hi] = hid_store[s * 2 * H + h];
    }
}

// ---------------------------------------------------------------------------
// positions_meta: the engine's gather/scatter position arrays straight from
// the device-side sample metadata — lp/tp (learning- and
// target-position rows of the (B, T+1, H) LSTM output), the inverse map
// feeding the dHext scatter, and per-sample lengths.  Replay-sampled
// batches have a different ragged layout every step, so the host-side
// numpy rebuild this replaces sat in the headline loop.
// ---------------------------------------------------------------------------
__global__ void positions_meta_kernel(
    const int* __restrict__ meta,   // (5, B): burn, learn, fwd, ...
    const int* __restrict__ seg,    // (B+1,)
    long* __restrict__ lp,          // (R,)
    long* __restrict__ tp,          // (R,)
    int* __restrict__ row_of,       // (B*T,) pre-filled with -1
    int* __restrict__ lens,         // (B,)
    int B, int T, int n, int max_learn) {
    int tid = blockIdx.x * blockDim.x + threadIdx.x;
    int b = tid / max_learn, j = tid % max_learn;
    if (b >= B) return;
    int burn = meta[b], learn = meta[B + b], fwd = meta[2 * B + b];
    if (j == 0) lens[b] = burn + learn + fwd;
    if (j < learn) {
        long r = seg[b] + j;
        int tl = burn + j;
        int tt = min(tl + n, burn + learn + fwd - 1);
        lp[r] = (long)b * (T + 1) + tl + 1;
        tp[r] = (long)b * (T + 1) + tt + 1;
        row_of[(long)b * T + tl] = (int)r;
    }
}

std::vector<torch::Tensor> positions_meta(torch::Tensor meta, torch::Tensor seg,
                                          int64_t T, int64_t n, int64_t R,
                                          int64_t max_learn) {
    TORCH_CHECK(meta.is_cuda() && meta.dtype() == torch::kInt32);
    int B = meta.size(1);
    auto i64 = meta.options().dtype(torch::kInt64);
    auto i32 = meta.options();
    auto lp = torch::empty({R}, i64);
    auto tp = torch::empty({R}, i64);
    auto row_of = torch::full({B * T}, -1, i32);
    auto lens = torch::empty({B}, i32);
    long total = (long)B * max_learn;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(positions_meta_kernel,
                       dim3((int)((total + 255) / 256)), dim3(256), 0,
                       stream.stream(), meta.data_ptr<int>(),
                       seg.data_ptr<int>(), lp.data_ptr<long>(),
                       tp.data_ptr<long>(), row_of.data_ptr<int>(),
                       lens.data_ptr<int>(), B, (int)T, (int)n,
                       (int)max_learn);
    return {lp, tp, row_of, lens};
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> replay_gather_meta(
    torch::Tensor idx, torch::Tensor burn_s, torch::Tensor learn_s,
    torch::Tensor fwd_s, torch::Tensor obs_start_s, torch::Tensor learn_off_s,
    int64_t spb) {
    int B = idx.size(0);
    TORCH_CHECK(B <= 1024);
    auto opts_i = idx.options().dtype(torch::kInt32);
    auto meta = torch::empty({5, B}, opts_i);
    auto seg = torch::empty({B + 1}, opts_i);
    int threads = 64;
    while (threads < B) threads *= 2;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(gather_meta_kernel, dim3(1), dim3(threads), 0,
                       stream.stream(), idx.data_ptr<long>(),
                       burn_s.data_ptr<int>(), learn_s.data_ptr<int>(),
                       fwd_s.data_ptr<int>(), obs_start_s.data_ptr<int>(),
                       learn_off_s.data_ptr<int>(), meta.data_ptr<int>(),
                       seg.data_ptr<int>(), B, (int)spb);
    return {meta, seg};
}

std::vector<torch::Tensor> replay_gather_batch(
    torch::Tensor obs_store, torch::Tensor la_store, torch::Tensor lr_store,
    torch::Tensor act_store, torch::Tensor nsr_store, torch::Tensor gam_store,
    torch::Tensor hid_store, torch::Tensor idx, torch::Tensor meta,
    torch::Tensor seg, torch::Tensor weights, int64_t T, int64_t A,
    int64_t max_learn, int64_t H, int64_t spb) {
    int B = idx.size(0);
    long obs_rows = obs_store.size(1);
    long frame_bytes = obs_store.size(2);
    long block_len = act_store.size(1);
    auto u8 = obs_store.options();
    auto f32 = lr_store.options();
    auto i64 = idx.options();
    auto obs_out = torch::empty({B, T, frame_bytes}, u8);
    auto la_out = torch::empty({B, T, A}, f32);
    auto lr_out = torch::empty({B, T}, f32);
    long Rmax = (long)B * max_learn;
    auto act_out = torch::empty({Rmax}, i64);
    auto nsr_out = torch::empty({Rmax}, f32);
    auto gam_out = torch::empty({Rmax}, f32);
    auto w_out = torch::empty({Rmax}, f32);
    auto hid_out = torch::empty({2, B, H}, f32);
    auto stream = at::cuda::getCurrentCUDAStream();

    hipLaunchKernelGGL(gather_frames_kernel, dim3(B * T), dim3(256), 0,
                       stream.stream(), obs_store.data_ptr<unsigned char>(),
                       la_store.data_ptr<unsigned char>(), lr_store.data_ptr<float>(),
                       meta.data_ptr<int>(), obs_out.data_ptr<unsigned char>(),
                       la_out.data_ptr<float>(), lr_out.data_ptr<float>(),
                       B, (int)T, (int)A, frame_bytes, obs_rows, (int)spb);

    int total = B * (int)max_learn;
    hipLaunchKernelGGL(gather_flat_kernel, dim3((total + 255) / 256), dim3(256),
                       0, stream.stream(), act_store.data_ptr<unsigned char>(),
                       nsr_store.data_ptr<float>(), gam_store.data_ptr<float>(),
                       hid_store.data_ptr<float>(), idx.data_ptr<long>(),
                       meta.data_ptr<int>(), seg.data_ptr<int>(),
                       weights.data_ptr<float>(), act_out.data_ptr<long>(),
                       nsr_out.data_ptr<float>(), gam_out.data_ptr<float>(),
                       w_out.data_ptr<float>(), hid_out.data_ptr<float>(),
                       B, (int)max_learn, (int)H, block_len, (int)spb);

    return {obs_out, la_out, lr_out, act_out, nsr_out, gam_out, w_out, hid_out};
}
