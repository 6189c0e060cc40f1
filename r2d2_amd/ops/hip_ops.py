"""Autograd-integrated wrappers around the gfx950 HIP extension.

Every wrapper asserts the extension is present when running on a CUDA/HIP
device — no silent eager fallback on GPU boxes (the eager/golden path is for
CPU and for tests, selected explicitly by the caller).
"""

from typing import Tuple

import torch

from . import _ext


def ext(required: bool = True):
    m = _ext.load(required=False)
    if m is None and required:
        raise RuntimeError(
            "r2d2_hip extension is not built; run __graft_entry__.build()")
    return m


def available() -> bool:
    return _ext.load(required=False) is not None


class _FusedDoubleQLoss(torch.autograd.Function):
    """loss, priorities = f(q_learn, ...) with dLoss/dq_learn computed in the
    forward sweep (single fused kernel: K9+K10+K11 of SURVEY.md §2.3)."""

    @staticmethod
    def forward(ctx, q_learn, q_online_tgt, q_target_tgt, action, n_step_reward,
                gamma_n, is_weights, seg_offsets, eps, kappa, loss_kind, eta):
        m = ext()
        loss, dq, abs_td, target = m.fused_double_q_loss(
            q_learn.contiguous().float(), q_online_tgt.contiguous().float(),
            q_target_tgt.contiguous().float(), action.contiguous(),
            n_step_reward.contiguous().float(), gamma_n.contiguous().float(),
            is_weights.contiguous().float(), eps, kappa,
            0 if loss_kind == "mse" else 1)
        prio = m.segment_priority(abs_td, seg_offsets, eta)
        ctx.save_for_backward(dq)
        ctx.mark_non_differentiable(prio)
        return loss.squeeze(0), prio

    @staticmethod
    def backward(ctx, grad_loss, _grad_prio):
        (dq,) = ctx.saved_tensors
        return (dq * grad_loss,) + (None,) * 11


def fused_double_q_loss(q_learn: torch.Tensor, q_online_tgt: torch.Tensor,
                        q_target_tgt: torch.Tensor, action: torch.Tensor,
                        n_step_reward: torch.Tensor, gamma_n: torch.Tensor,
                        is_weights: torch.Tensor, learning_steps: torch.Tensor,
                        eps: float = 1e-3, kappa: float = 1.0,
                        loss_kind: str = "huber",
                        eta: float = 0.9) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (scalar loss, per-sequence priorities (B,) on device)."""
    seg = torch.zeros(learning_steps.numel() + 1, dtype=torch.int32)
    seg[1:] = torch.cumsum(learning_steps.cpu().to(torch.int32), 0)
    seg = seg.to(q_learn.device, non_blocking=True)
    act = action.view(-1).long()
    return _FusedDoubleQLoss.apply(q_learn, q_online_tgt, q_target_tgt, act,
                                   n_step_reward, gamma_n, is_weights, seg,
                                   eps, kappa, loss_kind, eta)
