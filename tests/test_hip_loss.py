"""GPU numerics tests: fused double-Q loss kernel vs the fp32 eager golden."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops
    from r2d2_amd.ops import functional as Fn


def golden(q_learn, qo, qt, action, r, g, w, lengths, eps, kappa, kind, eta):
    target = Fn.double_q_target(qo, qt, r, g, eps)
    bq = q_learn.gather(1, action.view(-1, 1)).squeeze(1)
    loss = (w * Fn.per_step_loss(bq, target, kind, kappa)).mean()
    td = (bq - target).abs()
    prio = Fn.mixed_td_priority(td, lengths, eta)
    return loss, td, prio


@pytest.mark.parametrize("kind", ["huber", "mse"])
@pytest.mark.parametrize("A", [2, 9, 18])
def test_fused_loss_matches_golden(kind, A):
    torch.manual_seed(0)
    dev = "cuda"
    B, L = 16, 7
    lengths = torch.randint(1, L + 1, (B,))
    R = int(lengths.sum())
    q_learn = (torch.randn(R, A, device=dev) * 3).requires_grad_(True)
    qo = torch.randn(R, A, device=dev) * 2
    qt = torch.randn(R, A, device=dev) * 2
    action = torch.randint(0, A, (R,), device=dev)
    r = torch.randn(R, device=dev) * 5
    g = torch.rand(R, device=dev)
    w = torch.rand(R, device=dev) + 0.1

    loss, prio = hip_ops.fused_double_q_loss(
        q_learn, qo, qt, action, r, g, w, lengths,
        eps=1e-3, kappa=1.0, loss_kind=kind, eta=0.9)
    loss.backward()
    grad_hip = q_learn.grad.clone()

    q2 = q_learn.detach().clone().requires_grad_(True)
    loss_g, td_g, prio_g = golden(q2, qo, qt, action, r, g, w, lengths,
                                  1e-3, 1.0, kind, 0.9)
    loss_g.backward()

    assert torch.allclose(loss, loss_g, atol=1e-5, rtol=1e-5), (loss, loss_g)
    assert torch.allclose(prio, prio_g.to(dev), atol=1e-5)
    # fp32 target recomputation (sqrtf chain) differs by ULPs; the MSE grad
    # amplifies by 2*w*td, so compare with fp32-appropriate tolerance
    assert torch.allclose(grad_hip, q2.grad, atol=2e-5, rtol=1e-4)


def test_argmax_tie_break_first():
    """torch.argmax returns the first maximal index; the kernel must agree."""
    dev = "cuda"
    A = 5
    qo = torch.zeros(4, A, device=dev)     # all ties -> a* = 0
    qt = torch.arange(4 * A, device=dev, dtype=torch.float32).view(4, A)
    q_learn = torch.zeros(4, A, device=dev, requires_grad=True)
    action = torch.zeros(4, dtype=torch.long, device=dev)
    r = torch.zeros(4, device=dev)
    g = torch.ones(4, device=dev)
    w = torch.ones(4, device=dev)
    lengths = torch.tensor([2, 2])
    loss, _ = hip_ops.fused_double_q_loss(q_learn, qo, qt, action, r, g, w,
                                          lengths)
    target = Fn.value_rescale(Fn.inverse_value_rescale(qt[:, 0]))
    expect = (0.5 * target ** 2).mean()  # huber quadratic region? |td|<=1 not all
    td = -target
    hub = torch.where(td.abs() <= 1.0, 0.5 * td ** 2, td.abs() - 0.5)
    assert torch.allclose(loss, hub.mean(), atol=1e-5)


def test_extreme_values_value_rescale():
    dev = "cuda"
    x = torch.linspace(-500, 500, 64, device=dev)
    A = 3
    R = 64
    qo = torch.zeros(R, A, device=dev)
    qo[:, 1] = 1.0                      # a* = 1
    qt = torch.zeros(R, A, device=dev)
    qt[:, 1] = x
    q_learn = torch.zeros(R, A, device=dev, requires_grad=True)
    action = torch.ones(R, dtype=torch.long, device=dev)
    r = torch.zeros(R, device=dev)
    g = torch.ones(R, device=dev)
    w = torch.ones(R, device=dev)
    lengths = torch.tensor([64])
    loss, prio = hip_ops.fused_double_q_loss(q_learn, qo, qt, action, r, g, w,
                                             lengths)
    expect_target = Fn.value_rescale(Fn.inverse_value_rescale(x))
    assert torch.allclose(expect_target, x, atol=1e-2, rtol=1e-4)
    assert torch.isfinite(loss).all() and torch.isfinite(prio).all()


@pytest.mark.timeout(240)
def test_learner_hybrid_path_unsupported_config():
    """GPU configs outside the full engine's coverage (e.g. hidden_dim 256)
    run eager forward + the fused loss/priority kernels (Learner.train_step
    hybrid branch) — loss finite, priorities on-device."""
    import numpy as np
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner
    from bench import build_batch

    c = cfg.apply("mspacman", hidden_dim=256, batch_size=8, burn_in_steps=8,
                  learning_steps=8, forward_steps=3)
    torch.manual_seed(0)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    learner = Learner(None, None, model)
    learner.enable_hip_engine()
    assert learner.hip_engine and learner.engine is None
    batch = build_batch(c, torch.device("cuda:0"), seed=5)
    loss, prio = learner.train_step(batch)
    assert np.isfinite(float(loss))
    assert torch.is_tensor(prio) and prio.is_cuda and torch.isfinite(prio).all()
    cfg.apply("mspacman")
