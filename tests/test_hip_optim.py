"""GPU numerics: fused flat clip+Adam kernels vs torch clip_grad_norm_+Adam."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops


@pytest.mark.parametrize("max_norm,scale", [(40.0, 1.0), (0.5, 1.0), (40.0, 100.0)])
def test_fused_adam_matches_torch(max_norm, scale):
    torch.manual_seed(0)
    m = hip_ops.ext()
    n = 100_003
    dev = "cuda"
    p_ref = torch.randn(n, device=dev)
    p_hip = p_ref.clone()
    exp_avg = torch.zeros(n, device=dev)
    exp_avg_sq = torch.zeros(n, device=dev)
    norm_buf = torch.zeros(1, device=dev)

    ref_param = torch.nn.Parameter(p_ref.clone())
    opt = torch.optim.Adam([ref_param], lr=1e-3, eps=1e-3)

    for step in range(1, 6):
        g = torch.randn(n, device=dev) * scale
        # torch reference
        ref_param.grad = g.clone()
        torch.nn.utils.clip_grad_norm_([ref_param], max_norm)
        opt.step()
        # fused
        m.grad_sumsq(g, norm_buf)
        m.adam_step(p_hip, g, exp_avg, exp_avg_sq, norm_buf, max_norm,
                    1e-3, 0.9, 0.999, 1e-3, step)
    torch.cuda.synchronize()
    diff = (p_hip - ref_param.detach()).abs().max().item()
    assert diff < 1e-6, diff


def test_grad_sumsq_exact():
    m = hip_ops.ext()
    g = torch.randn(1_000_001, device="cuda")
    buf = torch.zeros(1, device="cuda")
    m.grad_sumsq(g, buf)
    torch.cuda.synchronize()
    ref = (g.double() ** 2).sum().item()
    assert abs(buf.item() - ref) / ref < 1e-5
