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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "r2d2_amd gfx950 HIP kernels";
    m.def("fused_double_q_loss", &fused_double_q_loss,
          "Fused double-Q target + TD + loss + dLoss/dQ (returns loss, dq, "
          "abs_td, target)");
    m.def("segment_priority", &segment_priority,
          "Per-sequence mixed max/mean |TD| priority over ragged segments");
}
