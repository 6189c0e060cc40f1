"""GPU numerics: persistent fused LSTM kernel vs an fp32 reference loop on
the same bf16-rounded inputs (length-masked, dual-network, BPTT)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops
    M_ = hip_ops.ext()

H = 512


def ref_lstm(X, Whh, h0, c0, lens):
    """fp32 reference with per-row length masking.  X: (B,T,4H) f32;
    returns Hs (B,T+1,H), Cs, and keeps graph for autograd."""
    B, T, _ = X.shape
    hs = [h0]
    cs = [c0]
    h, c = h0, c0
    for t in range(T):
        gates = X[:, t] + h @ Whh.t()
        i, f, g, o = gates.split(H, dim=1)
        i, f, g, o = torch.sigmoid(i), torch.sigmoid(f), torch.tanh(g), torch.sigmoid(o)
        c_new = f * c + i * g
        h_new = o * torch.tanh(c_new)
        mask = (torch.arange(t, t + 1, device=X.device).expand(B)
                < lens).float().unsqueeze(1)
        mask = (t < lens).float().unsqueeze(1)
        h = mask * h_new + (1 - mask) * h
        c = mask * c_new + (1 - mask) * c
        hs.append(h)
        cs.append(c)
    return torch.stack(hs, dim=1), torch.stack(cs, dim=1)


def make_inputs(B, T, seed=0, scale=0.5):
    torch.manual_seed(seed)
    dev = "cuda"
    X = (torch.randn(B, T, 4 * H, device=dev) * scale).bfloat16()
    Whh = (torch.randn(4 * H, H, device=dev) * (1.0 / np.sqrt(H))).bfloat16()
    h0 = torch.randn(B, H, device=dev) * 0.1
    c0 = torch.randn(B, H, device=dev) * 0.1
    lens = torch.randint(max(1, T - 6), T + 1, (B,), device=dev,
                         dtype=torch.int32)
    lens[0] = T
    return X, Whh, h0, c0, lens


def run_kernel_fwd(X, Whh, h0, c0, lens, want_stash=True):
    bar = torch.zeros(512, dtype=torch.int32, device="cuda")
    init = torch.stack([h0, c0]).contiguous()
    empty = torch.Tensor()
    outs = M_.lstm_fwd(X, empty, Whh, empty, init, empty, lens, bar, want_stash)
    return outs  # H0, C0, _, _, stash


@pytest.mark.parametrize("B,T", [(8, 12), (64, 20), (3, 5)])
def test_lstm_fwd_matches_ref(B, T):
    X, Whh, h0, c0, lens = make_inputs(B, T, seed=B + T)
    H0, C0, _, _, stash = run_kernel_fwd(X, Whh, h0, c0, lens)
    torch.cuda.synchronize()
    Hs, Cs = ref_lstm(X.float(), Whh.float(), h0, c0, lens)
    err_h = (H0.float() - Hs).abs().max().item()
    err_c = (C0 - Cs).abs().max().item()
    assert err_h < 3e-2, err_h
    assert err_c < 6e-2, err_c
    # masked rows frozen exactly
    for b in range(B):
        L = int(lens[b])
        if L < T:
            assert torch.equal(H0[b, L], H0[b, T])
            assert torch.equal(C0[b, L], C0[b, T])


def test_lstm_dual_network():
    B, T = 16, 10
    X0, Whh0, h00, c00, lens = make_inputs(B, T, seed=1)
    X1, Whh1, h01, c01, _ = make_inputs(B, T, seed=2)
    bar = torch.zeros(512, dtype=torch.int32, device="cuda")
    init0 = torch.stack([h00, c00]).contiguous()
    init1 = torch.stack([h01, c01]).contiguous()
    H0, C0, H1, C1, stash = M_.lstm_fwd(X0, X1, Whh0, Whh1, init0, init1,
                                        lens, bar, True)
    torch.cuda.synchronize()
    Hs0, _ = ref_lstm(X0.float(), Whh0.float(), h00, c00, lens)
    Hs1, _ = ref_lstm(X1.float(), Whh1.float(), h01, c01, lens)
    assert (H0.float() - Hs0).abs().max().item() < 3e-2
    assert (H1.float() - Hs1).abs().max().item() < 3e-2


def test_lstm_bwd_matches_autograd():
    B, T = 16, 12
    X, Whh, h0, c0, lens = make_inputs(B, T, seed=3, scale=0.3)
    H0, C0, _, _, stash = run_kernel_fwd(X, Whh, h0, c0, lens)
    dHext = (torch.randn(B, T, H, device="cuda") * 0.1)

    # kernel backward
    bar = torch.zeros(512, dtype=torch.int32, device="cuda")
    Whh_bwd = Whh.t().contiguous()  # (H, 4H)
    dg = M_.lstm_bwd(stash, C0, H0, dHext.contiguous(), Whh_bwd, lens, bar)
    torch.cuda.synchronize()

    # autograd reference (fp32, same rounded inputs); loss couples h_1..h_T
    X32 = X.float().requires_grad_(True)
    Whh32 = Whh.float().requires_grad_(True)
    Hs, Cs = ref_lstm(X32, Whh32, h0, c0, lens)
    loss = (Hs[:, 1:] * dHext).sum()
    loss.backward()

    # dgates == dL/dX (X enters gates additively)
    err = (dg.float() - X32.grad).abs().max().item()
    scale = X32.grad.abs().max().item()
    assert err < 3e-2 * max(1.0, scale), (err, scale)

    # dWhh via the wgrad GEMM over (B*T) rows: dgates^T @ h_prev
    h_prev = H0[:, :T].reshape(B * T, H).contiguous()
    dWhh, _ = M_.gemm_wgrad(dg.reshape(B * T, 4 * H).contiguous(),
                            torch.Tensor(), h_prev, False, False)
    errw = (dWhh - Whh32.grad).abs().max().item()
    scw = Whh32.grad.abs().max().item()
    assert errw < 5e-2 * max(1.0, scw), (errw, scw)
