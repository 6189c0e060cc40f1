// Python bindings for the r2d2_amd gfx950 HIP kernels.
#include <torch/extension.h>

#include <vector>

// loss_kernels.hip
std::vector<torch::Tensor> fused_double_q_loss(
    torch::Tensor q_learn, torch::Tensor q_online_tgt, torch::Tensor q_target_tgt,
    torch::Tensor action, torch::Tensor n_step_reward, torch::Tensor gamma_n,
    torch::Tensor is_weights, double eps, double kappa, int64_t loss_kind);
torch::Tensor segment_priority(torch::Tensor abs_td, torch::Tensor seg_offsets,
                               double eta);

// sumtree.hip
void sumtree_update(torch::Tensor tree, int64_t leaf_offset, torch::Tensor idxes,
                    torch::Tensor td, double alpha, int64_t old_ptr,
                    int64_t cur_ptr, int64_t seq_per_block, int64_t num_blocks);
std::vector<torch::Tensor> sumtree_sample(torch::Tensor tree, int64_t leaf_offset,
                                          int64_t num_levels, torch::Tensor jitter,
                                          int64_t n, double beta,
                                          int64_t max_idx);

// replay_gather.hip
std::vector<torch::Tensor> positions_meta(torch::Tensor meta, torch::Tensor seg,
                                          int64_t T, int64_t n, int64_t R,
                                          int64_t max_learn);
std::vector<torch::Tensor> replay_gather_meta(
    torch::Tensor idx, torch::Tensor burn_s, torch::Tensor learn_s,
    torch::Tensor fwd_s, torch::Tensor obs_start_s, torch::Tensor learn_off_s,
    int64_t spb);
std::vector<torch::Tensor> replay_gather_batch(
    torch::Tensor obs_store, torch::Tensor la_store, torch::Tensor lr_store,
    torch::Tensor act_store, torch::Tensor nsr_store, torch::Tensor gam_store,
    torch::Tensor hid_store, torch::Tensor idx, torch::Tensor meta,
    torch::Tensor seg, torch::Tensor weights, int64_t T, int64_t A,
    int64_t max_learn, int64_t H, int64_t spb);

// gemm_kernels.hip
torch::Tensor gemm_bias_act(torch::Tensor A, torch::Tensor Wt,
                            torch::Tensor bias, int64_t act, bool out_f32);
torch::Tensor gemm_dgrad(torch::Tensor dY, torch::Tensor act_out,
                         torch::Tensor W, bool relu_mask, int64_t k_out,
                         c10::optional<torch::Tensor> out_mask);
std::vector<torch::Tensor> gemm_wgrad(torch::Tensor dY, torch::Tensor act_out,
                                      torch::Tensor A, bool relu_mask,
                                      bool want_bias);
void gemm_wgrad_into(torch::Tensor dY, torch::Tensor act_out, torch::Tensor A,
                     bool relu_mask, torch::Tensor dW_out,
                     torch::Tensor db_out,
                     int64_t kd0, int64_t kd1, int64_t kd2);

// conv_kernels.hip
torch::Tensor conv_fwd(torch::Tensor in, torch::Tensor Wt, torch::Tensor bias,
                       int64_t conv_id, int64_t N, int64_t INH, int64_t INW,
                       int64_t OH, int64_t OW, bool relu);
torch::Tensor conv_dgrad(torch::Tensor dYp, torch::Tensor Wd, torch::Tensor taps,
                         int64_t N, int64_t PH, int64_t PW, int64_t COUT,
                         int64_t XH, int64_t XW, int64_t CIN,
                         int64_t y0, int64_t x0, int64_t S, int64_t pad,
                         torch::Tensor dX);
torch::Tensor conv_dgrad_dense(torch::Tensor dY, torch::Tensor Wd,
                               torch::Tensor taps, torch::Tensor actx,
                               int64_t N, int64_t OH, int64_t OW, int64_t COUT,
                               int64_t XH, int64_t XW, int64_t CIN,
                               int64_t y0, int64_t x0, int64_t S,
                               torch::Tensor dX);
std::vector<torch::Tensor> conv_wgrad(torch::Tensor dY, torch::Tensor act,
                                      torch::Tensor in, int64_t conv_id,
                                      int64_t N, int64_t INH, int64_t INW,
                                      int64_t OH, int64_t OW, int64_t COUT,
                                      int64_t K);
std::vector<torch::Tensor> conv_wgrad_band(torch::Tensor dY, torch::Tensor act,
                                           torch::Tensor in, int64_t conv_id,
                                           int64_t N);
torch::Tensor conv_fwd_band(torch::Tensor in, torch::Tensor Wt,
                            torch::Tensor bias, int64_t conv_id, int64_t N);

// lstm_kernels.hip
torch::Tensor assemble_rin(torch::Tensor latent, torch::Tensor la,
                           torch::Tensor lr, int64_t kin_pad);
void barrier_bench(torch::Tensor barrier_ws, int64_t steps, int64_t nblocks);
void handoff_bench(torch::Tensor barrier_ws, int64_t steps, int64_t nblocks);
void handoff_counter_bench(torch::Tensor barrier_ws, int64_t steps,
                           int64_t nblocks);
std::vector<torch::Tensor> lstm_fwd(
    torch::Tensor X0, torch::Tensor X1, torch::Tensor Whh0, torch::Tensor Whh1,
    torch::Tensor init0, torch::Tensor init1, torch::Tensor lens,
    torch::Tensor barrier_ws, bool want_stash);
torch::Tensor dueling_combine(torch::Tensor adv, torch::Tensor val, int64_t A);
torch::Tensor scatter_dh(torch::Tensor dh_a, torch::Tensor dh_v,
                         torch::Tensor row_of, int64_t BT);
std::vector<torch::Tensor> dueling_combine_bwd(torch::Tensor dq, int64_t PADA,
                                               int64_t PADV);
torch::Tensor lstm_bwd(torch::Tensor stash, torch::Tensor Cout,
                       torch::Tensor Hout, torch::Tensor dHext,
                       torch::Tensor Whh_bwd, torch::Tensor lens,
                       torch::Tensor barrier_ws);

// impala_kernels.hip
void conv3p_pool(torch::Tensor in, torch::Tensor Wt, torch::Tensor bias,
                 torch::Tensor pout, torch::Tensor parg, int64_t N,
                 int64_t stage);
void conv3p(torch::Tensor in, torch::Tensor Wt, torch::Tensor bias,
            torch::Tensor res, torch::Tensor mask, torch::Tensor out,
            int64_t N, int64_t H, int64_t W, bool relu_in, bool has_bias,
            int64_t epi);
std::vector<torch::Tensor> conv3p_wgrad(torch::Tensor dY, torch::Tensor in,
                                        int64_t N, int64_t H, int64_t W,
                                        bool relu_in);
void maxpool3s2_fwd(torch::Tensor in, torch::Tensor out, torch::Tensor arg,
                    int64_t N, int64_t H, int64_t W);
void maxpool3s2_bwd(torch::Tensor dOut, torch::Tensor arg, torch::Tensor dIn,
                    int64_t N, int64_t H, int64_t W, int64_t OH, int64_t OW);
void pack_frames(torch::Tensor frames, torch::Tensor out, int64_t H,
                 int64_t W);
torch::Tensor pad2dense(torch::Tensor in, int64_t N, int64_t H, int64_t W,
                        bool relu);
void dense2pad_mask(torch::Tensor dflat, torch::Tensor act_pad,
                    torch::Tensor out, int64_t N, int64_t H, int64_t W);

// optim_kernels.hip
void gather_pack(torch::Tensor flat, torch::Tensor m1, torch::Tensor m2,
                 torch::Tensor out);
torch::Tensor grad_sumsq(torch::Tensor grad, torch::Tensor norm_buf);
void adam_step(torch::Tensor param, torch::Tensor grad, torch::Tensor m,
               torch::Tensor v, torch::Tensor norm_buf, double max_norm,
               double lr, double beta1, double beta2, double eps,
               int64_t step);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "r2d2_amd gfx950 HIP kernels";
    m.def("fused_double_q_loss", &fused_double_q_loss,
          "Fused double-Q target + TD + loss + dLoss/dQ (returns loss, dq, "
          "abs_td, target)");
    m.def("segment_priority", &segment_priority,
          "Per-sequence mixed max/mean |TD| priority over ragged segments");
    m.def("sumtree_update", &sumtree_update,
          "GPU sum-tree priority update (duplicate-safe, ring-stale-masked)");
    m.def("sumtree_sample", &sumtree_sample,
          "GPU sum-tree stratified sample -> (idx, prio, is_weight)");
    m.def("replay_gather_meta", &replay_gather_meta,
          "Sampled-sequence metadata + segment offsets");
    m.def("positions_meta", &positions_meta,
          "engine gather/scatter position arrays from device-side sample "
          "metadata (no host round trip per ragged batch)");
    m.def("replay_gather_batch", &replay_gather_batch,
          "On-device padded batch assembly from the GPU block store");
    m.def("gemm_bias_act", &gemm_bias_act, "MFMA GEMM + bias + activation");
    m.def("gemm_dgrad", &gemm_dgrad,
          "MFMA GEMM backward-data (+ fused ReLU mask; optional dense "
          "k_out-column output; optional fused output ReLU mask)",
          pybind11::arg("dY"), pybind11::arg("act_out"), pybind11::arg("W"),
          pybind11::arg("relu_mask"), pybind11::arg("k_out") = 0,
          pybind11::arg("out_mask") = pybind11::none());
    m.def("gemm_wgrad", &gemm_wgrad, "MFMA GEMM backward-weight (+ bias grad)");
    m.def("gemm_wgrad_into", &gemm_wgrad_into,
          "GEMM backward-weight accumulated straight into .grad views "
          "(optional (d0,d1,d2)->(d2,d0,d1) k-axis store permutation)",
          pybind11::arg("dY"), pybind11::arg("act_out"), pybind11::arg("A"),
          pybind11::arg("relu_mask"), pybind11::arg("dW_out"),
          pybind11::arg("db_out"), pybind11::arg("kd0") = 0,
          pybind11::arg("kd1") = 0, pybind11::arg("kd2") = 0);
    m.def("conv_fwd", &conv_fwd, "Implicit-GEMM MFMA conv forward (NHWC)");
    m.def("conv_dgrad", &conv_dgrad, "MFMA conv backward-data (tap classes)");
    m.def("conv_dgrad_dense", &conv_dgrad_dense,
          "MFMA conv backward-data from dense pre-masked dY (bounds-checked "
          "taps, optional fused output ReLU mask)");
    m.def("scatter_dh", &scatter_dh,
          "dHext (BT, H) from head-backward rows via inverse position map");
    m.def("conv_wgrad", &conv_wgrad, "MFMA conv backward-weight");
    m.def("conv_wgrad_band", &conv_wgrad_band,
          "per-image band wgrad (whole input image LDS-staged, one dequant "
          "per element, register accumulation, one atomic flush per wg)");
    m.def("conv_fwd_band", &conv_fwd_band,
          "per-image band forward (image + weights LDS-resident)");
    m.def("lstm_fwd", &lstm_fwd,
          "Persistent fused LSTM forward (dual-network, length-masked)");
    m.def("lstm_bwd", &lstm_bwd, "Persistent fused LSTM BPTT backward");
    m.def("assemble_rin", &assemble_rin,
          "fused LSTM-input row assembly (latent | action | reward | pad)");
    m.def("barrier_bench", &barrier_bench, "grid barrier microbench");
    m.def("handoff_bench", &handoff_bench, "producer-flag handoff microbench");
    m.def("handoff_counter_bench", &handoff_counter_bench,
          "counter-aggregated handoff microbench");
    m.def("dueling_combine", &dueling_combine, "q = V + A - mean(A)");
    m.def("dueling_combine_bwd", &dueling_combine_bwd, "dueling combine backward");
    m.def("conv3p", &conv3p,
          "IMPALA 3x3 s1 p1 conv on halo-padded NHWC (fwd & dgrad-as-conv, "
          "fused relu-in / residual / mask epilogues)");
    m.def("conv3p_pool", &conv3p_pool,
          "IMPALA stage conv FUSED with maxpool 3x3 s2 (conv output stays "
          "in LDS; emits pooled activations + tap argmax)");
    m.def("conv3p_wgrad", &conv3p_wgrad,
          "IMPALA 3x3 conv backward-weight (padded dY, relu-in patches)");
    m.def("maxpool3s2_fwd", &maxpool3s2_fwd,
          "maxpool 3x3 s2 p1 forward + tap argmax");
    m.def("maxpool3s2_bwd", &maxpool3s2_bwd,
          "maxpool backward (atomic-free input-side gather)");
    m.def("pack_frames", &pack_frames,
          "u8 HWC frames -> halo-padded 8-channel u8");
    m.def("pad2dense", &pad2dense, "padded NHWC -> dense rows (+relu)");
    m.def("dense2pad_mask", &dense2pad_mask,
          "dense grad -> padded, masked by act>0 (relu backward)");
    m.def("gather_pack", &gather_pack,
          "regenerate prepacked weights from the flat param buffer "
          "(index-map gather, one launch per dtype)");
    m.def("grad_sumsq", &grad_sumsq, "flat gradient squared-norm reduction");
    m.def("adam_step", &adam_step,
          "fused multi-tensor clip + Adam on the flat parameter buffer");
}
