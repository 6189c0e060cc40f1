"""GPU numerics: MFMA GEMM and implicit-GEMM conv kernels vs fp32 eager
references (same bf16-rounded inputs, bf16-appropriate tolerances)."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops
    M_ = hip_ops.ext()


def close(a, b, rtol=3e-2, atol=None, name=""):
    a = a.float()
    b = b.float()
    if atol is None:
        atol = 3e-2 * max(1.0, float(b.abs().max()))
    ok = torch.allclose(a, b, rtol=rtol, atol=atol)
    if not ok:
        d = (a - b).abs()
        print(f"{name}: max diff {d.max().item()} at scale {b.abs().max().item()}")
    assert ok, name


# -------------------------------------------------------------------------
# GEMM
# -------------------------------------------------------------------------

def test_gemm_bias_act_matches_fp32():
    torch.manual_seed(0)
    M, K, N = 200, 96, 80
    A = torch.randn(M, K, device="cuda").bfloat16()
    W = torch.randn(N, K, device="cuda").bfloat16()
    b = torch.randn(N, device="cuda")
    out = M_.gemm_bias_act(A, W, b, 1, False)   # relu, bf16 out
    ref = F.relu(A.float() @ W.float().t() + b)
    close(out, ref, name="gemm relu bf16")
    out32 = M_.gemm_bias_act(A, W, b, 0, True)  # no act, f32 out
    ref32 = A.float() @ W.float().t() + b
    close(out32, ref32, name="gemm f32 out")
    # no bias
    out_nb = M_.gemm_bias_act(A, W, torch.empty(0, device="cuda"), 0, False)
    close(out_nb, A.float() @ W.float().t(), name="gemm nobias")


def test_gemm_asymmetric_transpose_detecting():
    """Asymmetric shapes+data catch operand/output transposes (guide G9)."""
    M, K, N = 64, 32, 64
    A = torch.zeros(M, K, device="cuda").bfloat16()
    A[3, 5] = 2.0
    W = torch.zeros(N, K, device="cuda").bfloat16()
    W[7, 5] = 3.0
    out = M_.gemm_bias_act(A, W, torch.empty(0, device="cuda"), 0, True)
    assert out[3, 7].item() == pytest.approx(6.0)
    assert out.abs().sum().item() == pytest.approx(6.0)


def test_gemm_dgrad_wgrad_match_autograd():
    torch.manual_seed(1)
    M, K, N = 300, 64, 96
    A = torch.randn(M, K, device="cuda").bfloat16()
    W = torch.randn(N, K, device="cuda").bfloat16()
    b = torch.randn(N, device="cuda")
    # forward with relu
    out = M_.gemm_bias_act(A, W, b, 1, False)
    dY = torch.randn(M, N, device="cuda").bfloat16()

    # reference in fp32 using the KERNEL's activation mask (the kernel masks
    # on its bf16 forward; an f32 forward flips boundary elements)
    mask = (out.float() > 0).float()
    dout = dY.float() * mask
    ref_dA = dout @ W.float()
    ref_dW = dout.t() @ A.float()
    ref_db = dout.sum(0)

    # dgrad: dA = (dY*mask) @ W  — kernel takes W stored (K, N)
    W_kn = W.t().contiguous()
    dA = M_.gemm_dgrad(dY, out, W_kn, True)
    close(dA, ref_dA, rtol=5e-2, name="dgrad")
    # wgrad
    dWt, db = M_.gemm_wgrad(dY, out, A, True, True)
    close(dWt, ref_dW, rtol=5e-2, atol=0.5, name="wgrad")
    close(db, ref_db, rtol=5e-2, atol=0.5, name="bgrad")


# -------------------------------------------------------------------------
# Conv forward
# -------------------------------------------------------------------------

def prepack_w(w):  # (COUT, CIN, KH, KW) -> (COUT, KH*KW*CIN)
    return w.permute(0, 2, 3, 1).reshape(w.shape[0], -1).contiguous()


@pytest.mark.parametrize("conv_id,cin,cout,k,s,inhw", [
    (1, 4, 32, 8, 4, 84),
    (2, 32, 64, 4, 2, 20),
    (3, 64, 64, 3, 1, 9),
])
def test_conv_fwd_matches_eager(conv_id, cin, cout, k, s, inhw):
    torch.manual_seed(conv_id)
    N = 7
    ohw = (inhw - k) // s + 1
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.2).bfloat16()
    b = torch.randn(cout, device="cuda") * 0.1
    if conv_id == 1:
        x_nchw = torch.randint(0, 256, (N, cin, inhw, inhw),
                               dtype=torch.uint8, device="cuda")
        x_in = x_nchw.permute(0, 2, 3, 1).contiguous()  # NHWC bytes
        ref_in = x_nchw.float() / 255.0
    else:
        x_nchw = (torch.randn(N, cin, inhw, inhw, device="cuda")).bfloat16()
        x_in = x_nchw.permute(0, 2, 3, 1).contiguous()
        ref_in = x_nchw.float()
    out = M_.conv_fwd(x_in, prepack_w(w), b, conv_id, N, inhw, inhw,
                      ohw, ohw, True)
    ref = F.relu(F.conv2d(ref_in, w.float(), b, stride=s))
    ref_nhwc = ref.permute(0, 2, 3, 1).reshape(N * ohw * ohw, cout)
    close(out, ref_nhwc, rtol=5e-2, name=f"conv{conv_id} fwd")


# -------------------------------------------------------------------------
# Conv backward (dgrad via tap classes, wgrad)
# -------------------------------------------------------------------------

def test_conv3_dgrad_s1():
    """3x3 stride-1 dgrad via padded upstream + 9 uniform taps."""
    torch.manual_seed(3)
    N, CIN, COUT, K, XH = 5, 64, 64, 3, 9
    OH = XH - K + 1  # 7
    w = (torch.randn(COUT, CIN, K, K, device="cuda") * 0.2).bfloat16()
    x32 = torch.randn(N, CIN, XH, XH, device="cuda", requires_grad=True)
    out = F.conv2d(x32, w.float(), stride=1)
    dy32 = torch.randn_like(out)
    out.backward(dy32)

    dy = dy32.bfloat16()
    pad = K - 1
    PH = OH + 2 * pad
    dyp = torch.zeros(N, PH, PH, COUT, device="cuda").bfloat16()
    dyp[:, pad:pad + OH, pad:pad + OH] = dy.permute(0, 2, 3, 1)
    # Wd[ci][t*COUT+co] = w[co, ci, dy, dx] with t = dy*K+dx ; index into
    # dyp at (y - dy + pad) -> taps (dy, dx)
    Wd = w.permute(1, 2, 3, 0).reshape(CIN, K * K * COUT).contiguous()
    # reorder to (CIN, t, COUT) with t-major: already (ci, dy, dx, co) ✓
    taps = torch.tensor([[dy, dx] for dy in range(K) for dx in range(K)],
                        dtype=torch.int32, device="cuda")
    dX = torch.zeros(N, XH, XH, CIN, device="cuda").bfloat16()
    M_.conv_dgrad(dyp, Wd, taps, N, PH, PH, COUT, XH, XH, CIN, 0, 0, 1, pad, dX)
    ref = x32.grad.permute(0, 2, 3, 1)
    close(dX, ref, rtol=5e-2, name="conv3 dgrad")


def test_conv2_dgrad_s2_parity_classes():
    torch.manual_seed(2)
    N, CIN, COUT, K, S, XH = 4, 32, 64, 4, 2, 20
    OH = (XH - K) // S + 1  # 9
    w = (torch.randn(COUT, CIN, K, K, device="cuda") * 0.2).bfloat16()
    x32 = torch.randn(N, CIN, XH, XH, device="cuda", requires_grad=True)
    out = F.conv2d(x32, w.float(), stride=S)
    dy32 = torch.randn_like(out)
    out.backward(dy32)

    dy = dy32.bfloat16()
    pad = 1
    PH = OH + 2 * pad  # 11
    dyp = torch.zeros(N, PH, PH, COUT, device="cuda").bfloat16()
    dyp[:, pad:pad + OH, pad:pad + OH] = dy.permute(0, 2, 3, 1)
    dX = torch.zeros(N, XH, XH, CIN, device="cuda").bfloat16()
    w_nhwc = w.permute(0, 2, 3, 1)  # (co, dy, dx, ci)
    for py in range(S):
        for px in range(S):
            tap_list = [(dy_, dx_) for dy_ in range(py, K, S)
                        for dx_ in range(px, K, S)]
            taps = torch.tensor(tap_list, dtype=torch.int32, device="cuda")
            # Wd[ci][t*COUT + co] = w[co, dy_t, dx_t, ci]
            Wd = torch.stack([w_nhwc[:, d, x_, :] for d, x_ in tap_list], dim=0)
            Wd = Wd.permute(2, 0, 1).reshape(CIN, len(tap_list) * COUT).contiguous()
            M_.conv_dgrad(dyp, Wd, taps, N, PH, PH, COUT, XH, XH, CIN,
                          py, px, S, pad, dX)
    ref = x32.grad.permute(0, 2, 3, 1)
    close(dX, ref, rtol=5e-2, name="conv2 dgrad")


@pytest.mark.parametrize("conv_id,cin,cout,k,s,inhw", [
    (1, 4, 32, 8, 4, 84),
    (2, 32, 64, 4, 2, 20),
    (3, 64, 64, 3, 1, 9),
])
def test_conv_wgrad(conv_id, cin, cout, k, s, inhw):
    torch.manual_seed(10 + conv_id)
    N = 6
    ohw = (inhw - k) // s + 1
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.2).bfloat16()
    b = torch.zeros(cout, device="cuda")
    if conv_id == 1:
        x_nchw = torch.randint(0, 256, (N, cin, inhw, inhw),
                               dtype=torch.uint8, device="cuda")
        x_in = x_nchw.permute(0, 2, 3, 1).contiguous()
        # the kernel dequantizes to bf16; round the reference the same way
        ref_in = (x_nchw.float() / 255.0).bfloat16().float()
    else:
        xb = torch.randn(N, cin, inhw, inhw, device="cuda").bfloat16()
        x_in = xb.permute(0, 2, 3, 1).contiguous()
        ref_in = xb.float()
    # forward (relu) then wgrad of loss sum(dY * out)
    out = M_.conv_fwd(x_in, prepack_w(w), b, conv_id, N, inhw, inhw,
                      ohw, ohw, True)
    dY = (torch.randn(N * ohw * ohw, cout, device="cuda") * 0.5).bfloat16()

    w32 = w.float().requires_grad_(True)
    b32 = b.clone().requires_grad_(True)
    out32 = F.relu(F.conv2d(ref_in, w32, b32, stride=s))
    dY_nchw = dY.reshape(N, ohw, ohw, cout).permute(0, 3, 1, 2).float()
    out32.backward(dY_nchw)

    dWt, db = M_.conv_wgrad(dY, out, x_in, conv_id, N, inhw, inhw, ohw, ohw,
                            cout, k * k * cin)
    ref_dw = prepack_w(w32.grad)  # (COUT, K) in (ky,kx,c) order, f32
    close(dWt, ref_dw, rtol=5e-2, atol=0.5, name=f"conv{conv_id} wgrad")
    close(db, b32.grad, rtol=5e-2, atol=0.5, name=f"conv{conv_id} bgrad")


# -------------------------------------------------------------------------
# Dense (unpadded) dgrad with fused output mask; gemm output mask; scatter
# -------------------------------------------------------------------------

def test_conv3_dgrad_dense_with_out_mask():
    """Bounds-checked dense dgrad == padded-staging dgrad; the output mask
    fuses the next conv's ReLU backward into the dX store."""
    torch.manual_seed(3)
    N, CIN, COUT, K, XH = 5, 64, 64, 3, 9
    OH = XH - K + 1  # 7
    w = (torch.randn(COUT, CIN, K, K, device="cuda") * 0.2).bfloat16()
    x32 = torch.randn(N, CIN, XH, XH, device="cuda", requires_grad=True)
    out = F.conv2d(x32, w.float(), stride=1)
    dy32 = torch.randn_like(out)
    out.backward(dy32)

    dy_dense = dy32.bfloat16().permute(0, 2, 3, 1).contiguous()  # (N,OH,OW,CO)
    Wd = w.permute(1, 2, 3, 0).reshape(CIN, K * K * COUT).contiguous()
    taps = torch.tensor([[dy, dx] for dy in range(K) for dx in range(K)],
                        dtype=torch.int32, device="cuda")
    dX = torch.zeros(N, XH, XH, CIN, device="cuda").bfloat16()
    M_.conv_dgrad_dense(dy_dense.view(-1, COUT), Wd, taps,
                        torch.empty(0, device="cuda"),
                        N, OH, OH, COUT, XH, XH, CIN, 0, 0, 1, dX)
    ref = x32.grad.permute(0, 2, 3, 1)
    close(dX, ref, rtol=5e-2, name="conv3 dgrad dense")

    # with output mask: dX masked by (actx > 0) at the store
    actx = torch.randn(N, XH, XH, CIN, device="cuda").bfloat16()
    dXm = torch.zeros_like(dX)
    M_.conv_dgrad_dense(dy_dense.view(-1, COUT), Wd, taps, actx,
                        N, OH, OH, COUT, XH, XH, CIN, 0, 0, 1, dXm)
    refm = ref * (actx.float() > 0)
    close(dXm, refm, rtol=5e-2, name="conv3 dgrad dense masked")


def test_conv2_dgrad_dense_parity_classes():
    torch.manual_seed(2)
    N, CIN, COUT, K, S, XH = 4, 32, 64, 4, 2, 20
    OH = (XH - K) // S + 1  # 9
    w = (torch.randn(COUT, CIN, K, K, device="cuda") * 0.2).bfloat16()
    x32 = torch.randn(N, CIN, XH, XH, device="cuda", requires_grad=True)
    out = F.conv2d(x32, w.float(), stride=S)
    dy32 = torch.randn_like(out)
    out.backward(dy32)

    dy_dense = (dy32.bfloat16().permute(0, 2, 3, 1).contiguous()
                .view(-1, COUT))
    dX = torch.zeros(N, XH, XH, CIN, device="cuda").bfloat16()
    w_nhwc = w.permute(0, 2, 3, 1)
    for py in range(S):
        for px in range(S):
            tap_list = [(dy_, dx_) for dy_ in range(py, K, S)
                        for dx_ in range(px, K, S)]
            taps = torch.tensor(tap_list, dtype=torch.int32, device="cuda")
            Wd = torch.stack([w_nhwc[:, d, x_, :] for d, x_ in tap_list], dim=0)
            Wd = (Wd.permute(2, 0, 1)
                  .reshape(CIN, len(tap_list) * COUT).contiguous())
            M_.conv_dgrad_dense(dy_dense, Wd, taps,
                                torch.empty(0, device="cuda"),
                                N, OH, OH, COUT, XH, XH, CIN, py, px, S, dX)
    ref = x32.grad.permute(0, 2, 3, 1)
    close(dX, ref, rtol=5e-2, name="conv2 dgrad dense")


def test_gemm_dgrad_out_mask():
    torch.manual_seed(4)
    M, K, N = 128, 96, 64
    dY = torch.randn(M, N, device="cuda").bfloat16()
    W_kn = torch.randn(K, N, device="cuda").bfloat16()
    om = torch.randn(M, K, device="cuda").bfloat16()
    dA = M_.gemm_dgrad(dY, torch.empty(0, device="cuda"), W_kn, False, 0, om)
    ref = (dY.float() @ W_kn.float().t()) * (om.float() > 0)
    close(dA, ref, rtol=5e-2, name="gemm dgrad out-mask")


def test_scatter_dh_matches_index_add():
    torch.manual_seed(5)
    B, T, H, R = 7, 11, 64, 40
    dh_a = torch.randn(R, H, device="cuda").bfloat16()
    dh_v = torch.randn(R, H, device="cuda").bfloat16()
    # unique positions
    pos = torch.randperm(B * T, device="cuda")[:R].to(torch.int64)
    row_of = torch.full((B * T,), -1, dtype=torch.int32, device="cuda")
    row_of[pos] = torch.arange(R, dtype=torch.int32, device="cuda")
    out = M_.scatter_dh(dh_a, dh_v, row_of, B * T)
    ref = torch.zeros(B * T, H, device="cuda")
    ref.index_add_(0, pos, dh_a.float() + dh_v.float())
    close(out, ref, rtol=1e-3, atol=1e-3, name="scatter_dh")


@pytest.mark.parametrize("conv_id,cin,cout,k,s,inhw", [
    (1, 4, 32, 8, 4, 84),
    (2, 32, 64, 4, 2, 20),
    (3, 64, 64, 3, 1, 9),
])
def test_conv_wgrad_band_matches_chunked(conv_id, cin, cout, k, s, inhw):
    """Per-image band wgrad == the chunked global-patch wgrad."""
    torch.manual_seed(20 + conv_id)
    N = 6
    ohw = (inhw - k) // s + 1
    if conv_id == 1:
        x_in = torch.randint(0, 256, (N, inhw, inhw, cin),
                             dtype=torch.uint8, device="cuda")
    else:
        x_in = torch.randn(N, inhw, inhw, cin, device="cuda").bfloat16()
    dY = (torch.randn(N * ohw * ohw, cout, device="cuda") * 0.5).bfloat16()
    act = torch.randn(N * ohw * ohw, cout, device="cuda").bfloat16()

    dWt_ref, db_ref = M_.conv_wgrad(dY, act, x_in, conv_id, N, inhw, inhw,
                                    ohw, ohw, cout, k * k * cin)
    dWt, db = M_.conv_wgrad_band(dY, act, x_in, conv_id, N)
    close(dWt, dWt_ref, rtol=2e-3, atol=1e-2, name=f"band wgrad {conv_id}")
    close(db, db_ref, rtol=2e-3, atol=1e-2, name=f"band bgrad {conv_id}")


@pytest.mark.parametrize("conv_id", [1, 2, 3])
def test_conv_fwd_band_matches_classic(conv_id):
    """Per-image band forward == the classic global-patch forward."""
    geo = {1: (4, 32, 8, 4, 84), 2: (32, 64, 4, 2, 20), 3: (64, 64, 3, 1, 9)}
    cin, cout, k, s, inhw = geo[conv_id]
    torch.manual_seed(40 + conv_id)
    N = 9
    ohw = (inhw - k) // s + 1
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.2).bfloat16()
    b = torch.randn(cout, device="cuda") * 0.1
    if conv_id == 1:
        x_in = torch.randint(0, 256, (N, inhw, inhw, cin),
                             dtype=torch.uint8, device="cuda")
    else:
        x_in = torch.randn(N, inhw, inhw, cin, device="cuda").bfloat16()
    ref = M_.conv_fwd(x_in, prepack_w(w), b, conv_id, N, inhw, inhw,
                      ohw, ohw, True)
    out = M_.conv_fwd_band(x_in, prepack_w(w), b, conv_id, N)
    assert torch.equal(out, ref), (out.float() - ref.float()).abs().max()


def test_gemm_fat_tile_matches_small():
    """The 128x128 fat-tile GEMM == the 64x64 kernel bitwise (same per-
    element MFMA accumulation chain)."""
    torch.manual_seed(6)
    M, K, N = 1100, 96, 160   # crosses the fat-tile dispatch threshold
    A = torch.randn(M, K, device="cuda").bfloat16()
    W = torch.randn(N, K, device="cuda").bfloat16()
    b = torch.randn(N, device="cuda")
    fat = M_.gemm_bias_act(A, W, b, 1, False)          # fat path (M>=1024)
    small = M_.gemm_bias_act(A[:1000], W, b, 1, False)  # small path
    assert torch.equal(fat[:1000], small)
    ref = torch.nn.functional.relu(A.float() @ W.float().t() + b)
    close(fat, ref, rtol=5e-2, name="fat gemm")
