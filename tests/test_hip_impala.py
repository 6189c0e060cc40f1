"""GPU numerics: IMPALA encoder kernels (ops/hip/impala_kernels.hip) vs fp32
eager references — halo-padded 3x3 convs (fwd/dgrad/wgrad, fused epilogues),
maxpool 3x3 s2 p1 fwd/bwd, frame packing, and the full encoder fwd+bwd vs
autograd through the eager ImpalaCNN (models/encoders.py)."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops
    from r2d2_amd.ops import impala as imp
    M_ = hip_ops.ext()

EMPTY = torch.Tensor()


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


def pad_nhwc(x):
    """(N, H, W, C) -> zero-halo (N, H+2, W+2, C)."""
    return F.pad(x, (0, 0, 1, 1, 1, 1))


def conv_ref(xf, w, b=None):
    """fp32 3x3 s1 p1 conv on NHWC input; returns NHWC."""
    y = F.conv2d(xf.permute(0, 3, 1, 2), w, b, stride=1, padding=1)
    return y.permute(0, 2, 3, 1)


def test_conv3p_fwd_matches_fp32():
    torch.manual_seed(0)
    for cin, cout, H in ((16, 16, 21), (16, 32, 12), (32, 32, 11), (8, 16, 10)):
        N = 5
        x = torch.randn(N, H, H, cin, device="cuda").bfloat16()
        w = torch.randn(cout, cin, 3, 3, device="cuda") * 0.2
        b = torch.randn(cout, device="cuda")
        xp = pad_nhwc(x).contiguous()
        out = torch.zeros(N, H + 2, H + 2, cout, device="cuda",
                          dtype=torch.bfloat16)
        wt = imp._pack_fwd(w, cin, "cuda")
        M_.conv3p(xp, wt, b, EMPTY, EMPTY, out, N, H, H, False, True, 0)
        ref = conv_ref(x.float(), w, b)
        close(out[:, 1:H + 1, 1:H + 1], ref, name=f"conv3p {cin}->{cout}")
        # halo untouched
        assert out[:, 0].abs().sum().item() == 0


def test_conv3p_relu_in_and_residual_epilogue():
    torch.manual_seed(1)
    N, H, C = 4, 14, 16
    x = torch.randn(N, H, H, C, device="cuda").bfloat16()
    w = torch.randn(C, C, 3, 3, device="cuda") * 0.2
    b = torch.randn(C, device="cuda")
    res = torch.randn(N, H, H, C, device="cuda").bfloat16()
    xp, resp = pad_nhwc(x).contiguous(), pad_nhwc(res).contiguous()
    out = torch.zeros_like(xp)
    wt = imp._pack_fwd(w, C, "cuda")
    M_.conv3p(xp, wt, b, resp, EMPTY, out, N, H, H, True, True, 1)
    ref = conv_ref(F.relu(x.float()), w, b) + res.float()
    close(out[:, 1:H + 1, 1:H + 1], ref, name="conv3p relu_in + residual")


def test_conv3p_mask_epilogues():
    torch.manual_seed(2)
    N, H, C = 3, 9, 32
    dy = torch.randn(N, H, H, C, device="cuda").bfloat16()
    w = torch.randn(C, C, 3, 3, device="cuda") * 0.2
    mask = torch.randn(N, H, H, C, device="cuda").bfloat16()
    res = torch.randn(N, H, H, C, device="cuda").bfloat16()
    dyp, mp, rp = (pad_nhwc(t).contiguous() for t in (dy, mask, res))
    wt = imp._pack_fwd(w, C, "cuda")
    out = torch.zeros_like(dyp)
    M_.conv3p(dyp, wt, EMPTY, EMPTY, mp, out, N, H, H, False, False, 2)
    ref = conv_ref(dy.float(), w) * (mask.float() > 0)
    close(out[:, 1:H + 1, 1:H + 1], ref, name="conv3p mask epi2")
    out3 = torch.zeros_like(dyp)
    M_.conv3p(dyp, wt, EMPTY, rp, mp, out3, N, H, H, False, False, 3)
    close(out3[:, 1:H + 1, 1:H + 1], ref + res.float(), name="conv3p epi3")


def test_conv3p_dgrad_via_flipped_weights():
    """dgrad of conv3x3 s1 p1 == conv3p over padded dY with _pack_dgrad."""
    torch.manual_seed(3)
    N, H, cin, cout = 4, 12, 16, 32
    x = torch.randn(N, H, H, cin, device="cuda", requires_grad=True)
    w = torch.randn(cout, cin, 3, 3, device="cuda") * 0.2
    y = F.conv2d(x.permute(0, 3, 1, 2), w, None, 1, 1)
    dy = torch.randn_like(y)
    y.backward(dy)
    dy_nhwc = dy.permute(0, 2, 3, 1).bfloat16()
    dyp = pad_nhwc(dy_nhwc).contiguous()
    wd = imp._pack_dgrad(w, "cuda")
    dx = torch.zeros(N, H + 2, H + 2, cin, device="cuda", dtype=torch.bfloat16)
    M_.conv3p(dyp, wd, EMPTY, EMPTY, EMPTY, dx, N, H, H, False, False, 0)
    close(dx[:, 1:H + 1, 1:H + 1], x.grad, name="conv3p dgrad")


def test_conv3p_wgrad_matches_autograd():
    torch.manual_seed(4)
    N, H, cin, cout = 6, 11, 32, 32
    xb = torch.randn(N, H, H, cin, device="cuda").bfloat16()
    w = torch.randn(cout, cin, 3, 3, device="cuda", requires_grad=True)
    x = xb.float().permute(0, 3, 1, 2).detach()
    y = F.conv2d(F.relu(x), w, None, 1, 1)
    dy = torch.randn_like(y)
    y.backward(dy)
    dyb = dy.permute(0, 2, 3, 1).bfloat16()
    dWt, db = M_.conv3p_wgrad(pad_nhwc(dyb).contiguous(),
                              pad_nhwc(xb).contiguous(), N, H, H, True)
    dW = imp.self_conv_grad(dWt, cout, cin)
    close(dW, w.grad, rtol=4e-2, name="conv3p wgrad relu_in")
    close(db, dy.sum((0, 2, 3)), name="conv3p wgrad bias")


def test_conv3p_wgrad_u8_frames():
    """The stage-0 wgrad path: u8 frames (8 padded channels), no relu-in."""
    torch.manual_seed(9)
    N, H = 4, 84
    frames = torch.randint(0, 256, (N, H, H, 4), device="cuda",
                           dtype=torch.uint8)
    xp = torch.zeros(N, H + 2, H + 2, 8, device="cuda", dtype=torch.uint8)
    M_.pack_frames(frames, xp, H, H)
    w = torch.randn(16, 4, 3, 3, device="cuda", requires_grad=True)
    x = (frames.float() / 255.0).permute(0, 3, 1, 2).detach()
    y = F.conv2d(x, w, None, 1, 1)
    dy = torch.randn_like(y)
    y.backward(dy)
    dyb = dy.permute(0, 2, 3, 1).bfloat16()
    dWt, db = M_.conv3p_wgrad(pad_nhwc(dyb).contiguous(), xp, N, H, H, False)
    dW = imp.self_conv_grad(dWt, 16, 4)
    close(dW, w.grad, rtol=4e-2, name="conv3p wgrad u8")
    close(db, dy.sum((0, 2, 3)), name="conv3p wgrad u8 bias")


def test_maxpool3s2_fwd_bwd():
    torch.manual_seed(5)
    for H in (84, 42, 21):
        N, C = 3, 16
        x = torch.randn(N, H, H, C, device="cuda").bfloat16()
        xp = pad_nhwc(x).contiguous()
        OH = (H + 1) // 2
        out = torch.zeros(N, OH + 2, OH + 2, C, device="cuda",
                          dtype=torch.bfloat16)
        arg = torch.empty(N, OH, OH, C, device="cuda", dtype=torch.uint8)
        M_.maxpool3s2_fwd(xp, out, arg, N, H, H)
        xt = x.float().permute(0, 3, 1, 2).requires_grad_()
        ref = F.max_pool2d(xt, 3, 2, 1)
        assert OH == ref.shape[-1]
        close(out[:, 1:OH + 1, 1:OH + 1],
              ref.detach().permute(0, 2, 3, 1), name=f"maxpool fwd {H}")
        dout = torch.randn_like(ref)
        ref.backward(dout)
        dop = pad_nhwc(dout.permute(0, 2, 3, 1).bfloat16()).contiguous()
        din = torch.zeros(N, H + 2, H + 2, C, device="cuda",
                          dtype=torch.bfloat16)
        M_.maxpool3s2_bwd(dop, arg, din, N, H, H, OH, OH)
        # ties: torch picks one winner, ours picks first tap — with random
        # float inputs ties are measure-zero, so compare exactly
        close(din[:, 1:H + 1, 1:H + 1],
              xt.grad.permute(0, 2, 3, 1), name=f"maxpool bwd {H}")


def test_pack_and_pad_utils():
    torch.manual_seed(6)
    Mn = 4
    frames = torch.randint(0, 256, (Mn, 84, 84, 4), device="cuda",
                           dtype=torch.uint8)
    xp = torch.zeros(Mn, 86, 86, 8, device="cuda", dtype=torch.uint8)
    M_.pack_frames(frames, xp, 84, 84)
    assert xp.shape == (Mn, 86, 86, 8)
    assert torch.equal(xp[:, 1:85, 1:85, :4], frames)
    assert xp[:, :, :, 4:].abs().sum().item() == 0
    assert xp[:, 0].abs().sum().item() == 0

    x = torch.randn(Mn, 11, 11, 32, device="cuda").bfloat16()
    xpad = pad_nhwc(x).contiguous()
    flat = M_.pad2dense(xpad, Mn, 11, 11, True)
    close(flat, F.relu(x.float()).reshape(Mn, -1), name="pad2dense relu")

    g = torch.randn(Mn, 11 * 11 * 32, device="cuda").bfloat16()
    out = torch.zeros_like(xpad)
    M_.dense2pad_mask(g, xpad, out, Mn, 11, 11)
    ref = g.float().reshape(Mn, 11, 11, 32) * (x.float() > 0)
    close(out[:, 1:12, 1:12], ref, name="dense2pad mask")


def test_impala_encoder_full_fwd_bwd_vs_autograd():
    """Full encoder: HIP latent + all 16 weight grads vs an eager reference
    that rounds every intermediate to bf16 (matching the kernel datapath, so
    maxpool argmax tie-routing decisions agree and the comparison is tight;
    a pure-fp32 reference legitimately diverges 10-17%% on deep layers via
    tie misrouting alone)."""
    from r2d2_amd.models.encoders import ImpalaCNN
    torch.manual_seed(7)
    Mn = 12
    enc = ImpalaCNN(4, 512).cuda()
    for p_ in enc.parameters():
        p_.grad = torch.zeros_like(p_)
    obs = torch.randint(0, 256, (Mn, 84, 84, 4), device="cuda",
                        dtype=torch.uint8)

    pack = imp.ImpalaPack(enc, "cuda", with_bwd=True)
    latent, st = imp.encoder_fwd(M_, pack, obs, True)

    def bf(t):  # round to bf16, keep fp32 autograd flow
        return t.bfloat16().float()

    def conv(c, x):  # conv with bf16-rounded weights (kernel datapath)
        return F.conv2d(x, bf(c.weight), c.bias, 1, 1)

    def ref_forward(x):
        for stage in enc.stages:
            x = bf(conv(stage.conv, x))
            x = bf(F.max_pool2d(x, 3, stride=2, padding=1))
            for res in (stage.res1, stage.res2):
                y = bf(conv(res.conv1, bf(F.relu(x))))
                y = bf(conv(res.conv2, bf(F.relu(y))))
                x = bf(x + y)
        x = F.relu(x).flatten(1)
        return F.relu(F.linear(bf(x), bf(enc.fc.weight), enc.fc.bias))

    x_eager = bf(obs.float() / 255.0).permute(0, 3, 1, 2)
    ref_lat = ref_forward(x_eager)
    close(latent, ref_lat.detach(), rtol=4e-2,
          atol=4e-2 * float(ref_lat.detach().abs().max()), name="impala latent")

    dlat = torch.randn(Mn, 512, device="cuda")
    # kernel backward accumulates into the (pre-zeroed) module .grad views;
    # snapshot those, re-zero, then run the eager autograd reference.
    imp.encoder_bwd(M_, pack, st, dlat.bfloat16(), latent)
    kgrads = {n: p_.grad.clone() for n, p_ in enc.named_parameters()}
    for p_ in enc.parameters():
        p_.grad.zero_()
    ref_lat.backward(dlat)

    def rel_fro(a, b, tol, name):
        a, b = a.float().flatten(), b.float().flatten()
        err = (a - b).norm() / (b.norm() + 1e-8)
        assert err < tol, f"{name}: rel fro {err:.4f}"

    for n, p_ in enc.named_parameters():
        # stages 0/1 sit under the 84x84 and 42x42 maxpools: MFMA bf16
        # accumulation order vs the reference's fp32-accumulate-then-round
        # flips near-tie pool argmaxes, deterministically re-routing whole
        # gradient elements.  This chain test guards STRUCTURE (a transpose
        # or layout bug shows up as rel-fro ~1); elementwise numerics are
        # covered tightly by the isolated kernel tests above.
        tol = 0.12 if (n.startswith("stages.2") or n.startswith("fc")) \
            else 0.20
        rel_fro(kgrads[n], p_.grad, tol, n)


def test_impala_engine_train_step():
    """Learner.train_step through the HIP engine on the seaquest preset:
    finite loss/priorities and decreasing loss over repeated steps."""
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner
    from bench import build_batch

    c = cfg.apply("seaquest_impala", batch_size=8, burn_in_steps=8,
                  learning_steps=8, forward_steps=3)
    torch.manual_seed(0)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="impala",
                    forward_steps=c.forward_steps)
    learner = Learner(None, None, model)
    learner.enable_hip_engine()
    assert learner.engine is not None and learner.engine.impala
    batch = build_batch(c, torch.device("cuda:0"), seed=11)
    losses = []
    for _ in range(12):
        loss, prio = learner.train_step(batch)
        loss = float(loss)
        assert np.isfinite(loss)
        assert torch.isfinite(prio).all()
        losses.append(loss)
    assert losses[-1] < losses[0]
    cfg.apply("mspacman")


def test_res_block_bwd_isolated():
    """One residual block (C=16, the s0r1a config) fwd+bwd via the exact
    engine call pattern vs autograd — localizes accumulation vs kernel bugs."""
    torch.manual_seed(12)
    N, H, C = 6, 42, 16
    x = torch.randn(N, H, H, C, device="cuda").bfloat16()
    w1 = torch.randn(C, C, 3, 3, device="cuda", requires_grad=True)
    b1 = torch.randn(C, device="cuda", requires_grad=True)
    w2 = torch.randn(C, C, 3, 3, device="cuda", requires_grad=True)
    b2 = torch.randn(C, device="cuda", requires_grad=True)
    for t in (w1, w2):
        t.data *= 0.2

    xp = pad_nhwc(x).contiguous()
    wt1, wt2 = imp._pack_fwd(w1, C, "cuda"), imp._pack_fwd(w2, C, "cuda")
    wd1, wd2 = imp._pack_dgrad(w1, "cuda"), imp._pack_dgrad(w2, "cuda")
    y1 = torch.zeros_like(xp)
    M_.conv3p(xp, wt1, b1.detach(), EMPTY, EMPTY, y1, N, H, H, True, True, 0)
    out = torch.zeros_like(xp)
    M_.conv3p(y1, wt2, b2.detach(), xp, EMPTY, out, N, H, H, True, True, 1)

    xe = x.float().permute(0, 3, 1, 2).detach().requires_grad_()
    y1e = F.conv2d(F.relu(xe), w1, b1, 1, 1)
    oute = xe + F.conv2d(F.relu(y1e), w2, b2, 1, 1)
    close(out[:, 1:H + 1, 1:H + 1], oute.detach().permute(0, 2, 3, 1),
          name="resblock fwd")

    dout = torch.randn_like(oute)
    oute.backward(dout)
    doutp = pad_nhwc(dout.permute(0, 2, 3, 1).bfloat16()).contiguous()
    # engine call pattern (impala.py encoder_bwd res loop)
    dW2, db2g = M_.conv3p_wgrad(doutp, y1, N, H, H, True)
    dy1 = torch.zeros_like(xp)
    M_.conv3p(doutp, wd2, EMPTY, EMPTY, y1, dy1, N, H, H, False, False, 2)
    dW1, db1g = M_.conv3p_wgrad(dy1, xp, N, H, H, True)
    dx = torch.zeros_like(xp)
    M_.conv3p(dy1, wd1, EMPTY, doutp, xp, dx, N, H, H, False, False, 3)

    def rel(a, b):
        return float((a.float().flatten() - b.float().flatten()).norm()
                     / (b.float().flatten().norm() + 1e-8))

    assert rel(imp.self_conv_grad(dW2, C, C), w2.grad) < 0.04, "dW2"
    assert rel(db2g, b2.grad) < 0.04, "db2"
    assert rel(imp.self_conv_grad(dW1, C, C), w1.grad) < 0.05, "dW1"
    assert rel(db1g, b1.grad) < 0.05, "db1"
    assert rel(dx[:, 1:H + 1, 1:H + 1],
               xe.grad.permute(0, 2, 3, 1)) < 0.05, "dx"


@pytest.mark.parametrize("encoder", ["nature", "impala"])
def test_hip_inference_matches_eager(encoder):
    """K15 single-step inference path vs the eager Network.forward."""
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import AgentState, Network
    from r2d2_amd.ops.engine import HipInference

    c = cfg.apply("seaquest_impala" if encoder == "impala" else "mspacman")
    torch.manual_seed(3)
    net = Network(c.action_dim, c.obs_shape, 512, encoder=encoder,
                  forward_steps=c.forward_steps).cuda().eval()
    E = 48
    obs = torch.randint(0, 256, (E, 4, 84, 84), device="cuda",
                        dtype=torch.uint8)
    la = torch.zeros(E, c.action_dim, device="cuda")
    la[torch.arange(E), torch.randint(0, c.action_dim, (E,))] = 1.0
    lr = torch.randn(E, 1, device="cuda") * 0.1
    h = torch.randn(1, E, 512, device="cuda") * 0.1
    ccell = torch.randn(1, E, 512, device="cuda") * 0.1

    inf = HipInference(net, "cuda")
    q, (h1, c1) = inf.forward(obs, la, lr, (h, ccell))

    state = AgentState.__new__(AgentState)
    state.obs = obs
    state.action_dim = c.action_dim
    state.last_action = la
    state.last_reward = lr
    state.hidden_state = (h, ccell)
    with torch.no_grad():
        q_ref, (h_ref, c_ref) = net(state)

    scale = float(q_ref.abs().max())
    assert torch.allclose(q.float(), q_ref, atol=0.05 * max(1.0, scale)), \
        float((q.float() - q_ref).abs().max())
    assert torch.allclose(h1, h_ref, atol=0.03), \
        float((h1 - h_ref).abs().max())
    # greedy actions agree almost everywhere
    agree = (q.float().argmax(1) == q_ref.argmax(1)).float().mean()
    assert agree > 0.9
    cfg.apply("mspacman")


@pytest.mark.timeout(300)
def test_vector_actor_gpu_hip_inference():
    """VectorActor on cuda drives env ticks through the K15 HIP path and
    produces valid replay blocks."""
    import queue
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import Network
    from r2d2_amd.train import epsilon_ladder
    from r2d2_amd.worker import VectorActor

    c = cfg.apply("mspacman_gpu_replay", num_actors=8, block_length=40,
                  burn_in_steps=8, learning_steps=8, forward_steps=3,
                  actor_update_interval=64, max_episode_steps=200)
    torch.manual_seed(0)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    sq = queue.Queue()
    va = VectorActor(epsilon_ladder(8), model, [sq], device="cuda", seed=4)
    assert va.hip_inf is not None, "K15 path must be active on GPU"
    total = va.run(stop_after_steps=400)
    assert total >= 400
    n = 0
    while not sq.empty():
        block, prio, reward = sq.get()
        assert np.isfinite(prio).all()
        assert block.obs.dtype == np.uint8
        n += 1
    assert n >= 4
    cfg.apply("mspacman")


@pytest.mark.timeout(420)
def test_train_gpu_replay_topology():
    """configs[2] end-to-end on one GPU: VectorActor (K15 inference) feeds
    the learner process's GPU-resident replay; the learner trains through
    the HIP engine.  Run in-process with a queue and a driver thread."""
    import queue
    import threading
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import Network
    from r2d2_amd.train import epsilon_ladder
    from r2d2_amd.worker import Learner, VectorActor

    c = cfg.apply("mspacman_gpu_replay", num_actors=8, block_length=40,
                  burn_in_steps=8, learning_steps=8, forward_steps=3,
                  batch_size=16, buffer_capacity=8_000, learning_starts=400,
                  training_steps=25, max_episode_steps=300, log_interval=5,
                  actor_update_interval=64, save_interval=10_000)
    torch.manual_seed(0)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    model.share_memory()
    sq = queue.Queue()
    learner = Learner(None, None, model, model_dir="/tmp/r2d2_test_models")
    learner.enable_hip_engine()
    assert learner.engine is not None

    va = VectorActor(epsilon_ladder(8), model, [sq], device="cuda", seed=1)
    stop = threading.Event()

    def drive():
        while not stop.is_set():
            va.run(stop_after_steps=100)

    t = threading.Thread(target=drive)
    t.start()
    try:
        learner.run_with_gpu_replay([sq])
    finally:
        stop.set()
        t.join(timeout=60)   # join before interpreter teardown (a daemon
        torch.cuda.synchronize()  # thread mid-CUDA-call aborts at exit)
    assert learner.num_updates == c.training_steps
    cfg.apply("mspacman")


@pytest.mark.parametrize("stage,cin,cout,ht", [(0, 8, 16, 84),
                                               (1, 16, 32, 42),
                                               (2, 32, 32, 21)])
def test_conv3p_pool_matches_separate_ops(stage, cin, cout, ht):
    """Fused stage-conv+maxpool == conv3p followed by maxpool3s2_fwd
    (bit-exact: same bf16 conv values, same tap argmax tie-breaks)."""
    torch.manual_seed(30 + stage)
    N = 4
    pout_h = (ht + 1) // 2
    w = (torch.randn(cout, cin, 3, 3, device="cuda") * 0.2).bfloat16()
    b = torch.randn(cout, device="cuda") * 0.1
    wt = w.permute(0, 2, 3, 1).reshape(cout, -1).contiguous()
    if stage == 0:
        xin = torch.zeros(N, ht + 2, ht + 2, cin, dtype=torch.uint8,
                          device="cuda")
        xin[:, 1:-1, 1:-1] = torch.randint(
            0, 256, (N, ht, ht, cin), dtype=torch.uint8, device="cuda")
    else:
        xin = torch.zeros(N, ht + 2, ht + 2, cin, device="cuda").bfloat16()
        xin[:, 1:-1, 1:-1] = torch.randn(N, ht, ht, cin,
                                         device="cuda").bfloat16()

    # reference: separate kernels
    conv_out = torch.zeros(N, ht + 2, ht + 2, cout, device="cuda").bfloat16()
    M_.conv3p(xin, wt, b, EMPTY, EMPTY, conv_out, N, ht, ht, False, True, 0)
    pooled_ref = torch.zeros(N, pout_h + 2, pout_h + 2, cout,
                             device="cuda").bfloat16()
    arg_ref = torch.zeros(N, pout_h, pout_h, cout, dtype=torch.uint8,
                          device="cuda")
    M_.maxpool3s2_fwd(conv_out, pooled_ref, arg_ref, N, ht, ht)

    pooled = torch.zeros_like(pooled_ref)
    arg = torch.zeros_like(arg_ref)
    M_.conv3p_pool(xin, wt, b, pooled, arg, N, stage)
    assert torch.equal(pooled, pooled_ref), \
        (pooled - pooled_ref).abs().max()
    assert torch.equal(arg, arg_ref)
