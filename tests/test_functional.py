import numpy as np
import pytest
import torch

from r2d2_amd.ops import functional as Fn


def test_value_rescale_roundtrip():
    x = torch.linspace(-200, 200, 1001, dtype=torch.float64)
    y = Fn.inverse_value_rescale(Fn.value_rescale(x))
    assert torch.allclose(x, y, atol=1e-6)
    # h is monotone and odd
    h = Fn.value_rescale(x)
    assert (h[1:] > h[:-1]).all()
    assert torch.allclose(h, -Fn.value_rescale(-x), atol=1e-12)


def test_value_rescale_known_values():
    x = torch.tensor([0.0, 3.0, -3.0, 99.0])
    h = Fn.value_rescale(x, eps=1e-3)
    expect = torch.tensor([0.0, 1.003, -1.003, 9.099])
    assert torch.allclose(h, expect, atol=1e-4)


def test_n_step_return_matches_loop():
    rng = np.random.default_rng(0)
    r = rng.normal(size=37).astype(np.float32)
    n, gamma = 5, 0.997
    out = Fn.n_step_return(r, n, gamma)
    for t in range(len(r)):
        expect = sum(gamma ** i * r[t + i] for i in range(min(n, len(r) - t)))
        assert abs(out[t] - expect) < 1e-5, t


def test_n_step_return_matches_reference_convolve():
    """Same construction the reference uses (worker.py:466-469)."""
    rng = np.random.default_rng(1)
    r = list(rng.normal(size=23).astype(np.float32))
    n, gamma = 5, 0.99
    padded = r + [0.0] * (n - 1)
    kernel = [gamma ** (n - 1 - i) for i in range(n)]
    expect = np.convolve(padded, kernel, "valid").astype(np.float32)
    got = Fn.n_step_return(np.array(r, dtype=np.float32), n, gamma)
    np.testing.assert_allclose(got, expect, atol=1e-5)


@pytest.mark.parametrize("T,n", [(10, 5), (3, 5), (5, 5), (1, 5), (10, 1)])
def test_gamma_vector(T, n):
    g = 0.997
    mfs = min(T, n)
    # bootstrap cut: last mfs entries decay gamma^mfs..gamma^1
    v = Fn.gamma_vector(T, n, g, done=False)
    expect = [g ** n] * (T - mfs) + [g ** i for i in range(mfs, 0, -1)]
    np.testing.assert_allclose(v, np.array(expect, dtype=np.float32), rtol=1e-6)
    # terminal: last mfs entries are exactly 0
    v = Fn.gamma_vector(T, n, g, done=True)
    expect = [g ** n] * (T - mfs) + [0.0] * mfs
    np.testing.assert_allclose(v, np.array(expect, dtype=np.float32), rtol=1e-6)


def test_mixed_td_priority():
    td = np.array([1.0, 2.0, 3.0, 10.0, 0.0], dtype=np.float32)
    lengths = np.array([3, 2])
    out = Fn.mixed_td_priority_np(td, lengths, eta=0.9)
    assert abs(out[0] - (0.9 * 3 + 0.1 * 2)) < 1e-6
    assert abs(out[1] - (0.9 * 10 + 0.1 * 5)) < 1e-6
    # torch golden agrees
    out_t = Fn.mixed_td_priority(torch.from_numpy(td), torch.from_numpy(lengths), 0.9)
    np.testing.assert_allclose(out, out_t.numpy(), rtol=1e-6)


def test_per_step_loss():
    q = torch.tensor([0.0, 1.0, 5.0])
    t = torch.tensor([0.5, 1.0, 0.0])
    mse = Fn.per_step_loss(q, t, "mse")
    assert torch.allclose(mse, (q - t) ** 2)
    hub = Fn.per_step_loss(q, t, "huber", kappa=1.0)
    assert abs(hub[0].item() - 0.125) < 1e-6     # quadratic region
    assert abs(hub[2].item() - (5 - 0.5)) < 1e-6  # linear region


def test_double_q_target():
    torch.manual_seed(0)
    B, A = 16, 5
    qo = torch.randn(B, A)
    qt = torch.randn(B, A)
    r = torch.randn(B)
    g = torch.full((B,), 0.997 ** 5)
    out = Fn.double_q_target(qo, qt, r, g)
    a_star = qo.argmax(1)
    expect = Fn.value_rescale(
        r + g * Fn.inverse_value_rescale(qt[torch.arange(B), a_star]))
    assert torch.allclose(out, expect, atol=1e-6)


def test_mixed_td_priority_full_block_uint8_lengths():
    """Regression: uint8 segment lengths from a FULL 400-step block must not
    wrap the running offset at 256 (found live — the reference stores
    per-sequence learning steps as uint8, worker.py Block)."""
    import numpy as np
    from r2d2_amd.ops import functional as Fn

    lengths = np.full(10, 40, dtype=np.uint8)   # sums to 400 > 255
    abs_td = np.arange(400, dtype=np.float32)
    out = Fn.mixed_td_priority_np(abs_td, lengths, eta=0.9)
    assert out.shape == (10,)
    # last segment covers [360, 400): max 399, mean 379.5
    assert abs(out[-1] - (0.9 * 399 + 0.1 * 379.5)) < 1e-3
