"""GPU end-to-end: the full HIP engine (conv+LSTM+heads+loss, manual
backward) vs the eager fp32 golden learner on identical weights and batch."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd import config as cfg
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner


def make_learners(seed=0):
    c = cfg.apply("mspacman", batch_size=8, burn_in_steps=8, learning_steps=8,
                  forward_steps=3, amp=False, dtype="fp32")
    torch.manual_seed(seed)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    eager = Learner(None, None, model)
    hip = Learner(None, None, model)
    hip.enable_hip_engine()
    assert hip.engine is not None
    return c, eager, hip


def cos(a, b):
    a, b = a.flatten().float(), b.flatten().float()
    denom = a.norm() * b.norm()
    if denom == 0:
        return 1.0
    return float((a @ b) / denom)


def test_engine_matches_eager_loss_and_grads():
    c, eager, hip = make_learners()
    from bench import build_batch
    batch_a = build_batch(c, torch.device("cuda"), seed=5)
    import copy
    batch_b = copy.deepcopy(batch_a)

    # eager step (keep grads by stubbing the optimizer step)
    opt_e = eager.optimizer
    loss_e, prio_e = eager.train_step(batch_a)
    grads_e = {n: p.grad.clone() for n, p in eager.online_net.named_parameters()}

    # engine-only call (no optimizer step) so grads compare at equal weights
    batch_b.to(torch.device("cuda"))
    loss_h2, prio_h2 = hip.engine.train_step(batch_b)
    grads_h = {n: p.grad.clone() for n, p in hip.online_net.named_parameters()}

    lf_e, lf_h = float(loss_e), float(loss_h2)
    assert abs(lf_e - lf_h) < 0.05 * max(1.0, abs(lf_e)), (lf_e, lf_h)

    pe = prio_e if isinstance(prio_e, np.ndarray) else prio_e.cpu().numpy()
    ph = prio_h2.cpu().numpy()
    np.testing.assert_allclose(ph, pe, rtol=0.1, atol=0.05)

    bad = []
    for n, ge in grads_e.items():
        gh = grads_h[n]
        cs = cos(ge, gh)
        rel = float((gh.float() - ge.float()).norm() / (ge.float().norm() + 1e-8))
        if cs < 0.98 and rel > 0.1:
            bad.append((n, cs, rel))
    assert not bad, bad


def test_engine_training_reduces_loss():
    c, eager, hip = make_learners(seed=1)
    from bench import build_batch
    batch = build_batch(c, torch.device("cuda"), seed=9)
    first = last = None
    for i in range(20):
        loss, _ = hip.train_step(batch)
        if i == 0:
            first = float(loss)
        last = float(loss)
    assert np.isfinite(last)
    assert last < first, (first, last)


def test_weight_bus_roundtrip():
    """WeightBus publish/pull: the learner's packed online weights land
    bit-exactly in a separate inference pack (same-process; the live demo
    exercises the cross-process CUDA-IPC path)."""
    from r2d2_amd.parallel.weight_bus import WeightBus, pack_tensors
    from r2d2_amd.ops.engine import HipInference

    c, eager, hip = make_learners(seed=3)
    engine = hip.engine
    bus = WeightBus(pack_tensors(engine.target), "cuda")

    # a second network with different weights = the actor's inference pack
    torch.manual_seed(99)
    from r2d2_amd.models.network import Network
    other = Network(c.action_dim, c.obs_shape, c.hidden_dim,
                    encoder=c.encoder, forward_steps=c.forward_steps,
                    mlp_hidden=c.mlp_hidden).cuda()
    inf = HipInference(other, torch.device("cuda"), c)

    before = {n: t.clone() for n, t in pack_tensors(inf.pack).items()}
    bus.publish(engine.online)
    ver = bus.pull_into(inf.pack, 0)
    assert ver == 2   # seqlock: odd while publishing, even when stable

    src = pack_tensors(engine.online)
    dst = pack_tensors(inf.pack)
    changed = 0
    for name, t in dst.items():
        assert torch.equal(t, src[name].view(t.shape)), name
        if not torch.equal(t, before[name]):
            changed += 1
    assert changed > 0
    # unchanged version -> no-op
    assert bus.pull_into(inf.pack, ver) == ver


def test_fast_refresh_matches_python_repack():
    """The gather-kernel repack (bit-probed index maps) must reproduce the
    python repack exactly after arbitrary parameter changes."""
    from r2d2_amd.ops.engine import _NetPack, _pack_items, _item_get

    c, eager, hip = make_learners(seed=7)
    engine = hip.engine

    # perturb the flat params, fast-refresh, then compare against a fresh
    # python-packed copy of the same module
    with torch.no_grad():
        engine.flat_param.add_(
            torch.randn_like(engine.flat_param) * 0.01)
    engine.refresh_online()   # fast path (maps)
    ref = _NetPack(engine.online_net, engine.device, engine.A,
                   with_bwd=True)   # python repack of the same weights
    fast_items = {n: _item_get(h, a, k)
                  for n, h, a, k in _pack_items(engine.online)}
    ref_items = {n: _item_get(h, a, k)
                 for n, h, a, k in _pack_items(ref)}
    assert set(fast_items) == set(ref_items)
    for n, t in fast_items.items():
        assert torch.equal(t, ref_items[n].view(t.shape)), n


def build_ragged_batch(c, device, seed):
    """A batch with per-sample burn/learn/forward variation — the layout
    real LocalBuffer blocks produce at episode starts and tail cuts."""
    from r2d2_amd.worker import TrainingBatch

    rng = np.random.default_rng(seed)
    g = torch.Generator(device="cpu").manual_seed(seed)
    B = c.batch_size
    burn = torch.from_numpy(
        rng.integers(0, c.burn_in_steps + 1, B)).long()
    learn = torch.from_numpy(
        rng.integers(1, c.learning_steps + 1, B)).long()
    fwd = torch.from_numpy(
        rng.integers(1, c.forward_steps + 1, B)).long()
    T = int((burn + learn + fwd).max())
    A = c.action_dim
    R = int(learn.sum())
    obs = torch.randint(0, 256, (B, T) + tuple(c.obs_shape),
                        dtype=torch.uint8, generator=g)
    la = torch.zeros(B, T, A)
    la[torch.arange(B)[:, None], torch.arange(T)[None, :],
       torch.randint(0, A, (B, T), generator=g)] = 1.0
    lr = torch.randn(B, T, generator=g) * 0.1
    hidden = torch.randn(2, B, c.hidden_dim, generator=g) * 0.05
    batch = TrainingBatch(
        obs=obs, last_action=la, last_reward=lr, hidden=hidden,
        action=torch.randint(0, A, (R, 1), generator=g),
        n_step_reward=torch.randn(R, generator=g).abs(),
        gamma=torch.full((R,), c.gamma ** c.forward_steps),
        burn_in_steps=burn, learning_steps=learn, forward_steps=fwd,
        idxes=np.arange(B), is_weights=torch.rand(R, generator=g) * 0.5 + 0.5,
        old_ptr=0, env_steps=0)
    return batch.to(device, non_blocking=False)


def test_engine_matches_eager_on_ragged_layouts():
    """Variable burn-in/learning/forward lengths (episode starts, tail
    cuts) must agree with the eager golden — the positions/scatter/loss
    machinery is layout-dependent."""
    c, eager, hip = make_learners(seed=11)
    for trial in range(3):
        batch_a = build_ragged_batch(c, torch.device("cuda"), seed=50 + trial)
        batch_b = build_ragged_batch(c, torch.device("cuda"), seed=50 + trial)

        eager.optimizer.zero_grad(set_to_none=True)
        loss_e, prio_e = eager.train_step(batch_a)
        grads_e = {n: p.grad.clone()
                   for n, p in eager.online_net.named_parameters()}

        hip.engine.flat_grad.zero_()
        loss_h, prio_h = hip.engine.train_step(batch_b)
        grads_h = {n: p.grad.clone()
                   for n, p in hip.online_net.named_parameters()}

        lf_e, lf_h = float(loss_e), float(loss_h)
        assert abs(lf_e - lf_h) < 0.05 * max(1.0, abs(lf_e)), \
            (trial, lf_e, lf_h)
        pe = prio_e if isinstance(prio_e, np.ndarray) else prio_e.cpu().numpy()
        np.testing.assert_allclose(prio_h.cpu().numpy(), pe,
                                   rtol=0.15, atol=0.05)
        bad = []
        for n, ge in grads_e.items():
            gh = grads_h[n]
            cs = cos(ge, gh)
            rel = float((gh.float() - ge.float()).norm()
                        / (ge.float().norm() + 1e-8))
            if cs < 0.98 and rel > 0.12:
                bad.append((trial, n, cs, rel))
        assert not bad, bad


def test_engine_checkpoint_resume_roundtrip(tmp_path):
    """Save/resume on the HIP engine path: the 4-tuple + .train.pth sidecar
    must restore flat params AND the engine's fused-Adam moments so
    training continues bit-compatibly."""
    import os
    c, eager, hip = make_learners(seed=21)
    from bench import build_batch
    batch = build_batch(c, torch.device("cuda"), seed=31)
    for _ in range(3):
        loss, prio = hip.train_step(batch)
    hip.model_dir = str(tmp_path)
    hip.game_name = "Resume"
    hip.save(start_time=0.0)
    path = os.path.join(str(tmp_path), f"Resume{hip.num_updates}.pth")
    assert os.path.exists(path) and os.path.exists(
        path[:-4] + ".train.pth")

    _, _, hip2 = make_learners(seed=99)   # different init
    hip2.load_checkpoint(path)
    assert torch.equal(hip2.engine.flat_param, hip.engine.flat_param)
    assert torch.equal(hip2.engine.exp_avg, hip.engine.exp_avg)
    assert torch.equal(hip2.engine.exp_avg_sq, hip.engine.exp_avg_sq)
    assert hip2.engine.adam_t == hip.engine.adam_t
    assert hip2.num_updates == hip.num_updates

    l1, _ = hip.train_step(batch)
    l2, _ = hip2.train_step(batch)
    assert abs(float(l1) - float(l2)) < 1e-5, (float(l1), float(l2))


def test_weight_bus_concurrent_publish_pull_consistency():
    """Torn-read protection: a puller racing a publisher must always see an
    internally consistent snapshot (every tensor from the same publish
    generation — the version word re-check redoes overlapped pulls)."""
    import threading
    import types
    from r2d2_amd.parallel.weight_bus import WeightBus, pack_tensors

    def make_pack(val):
        return types.SimpleNamespace(
            a=torch.full((1000,), float(val), device="cuda",
                         dtype=torch.bfloat16),
            b=torch.full((64, 64), float(val), device="cuda",
                         dtype=torch.bfloat16),
            c=torch.full((333,), float(val), device="cuda",
                         dtype=torch.float32))

    src = make_pack(0)
    dst = make_pack(-1)
    bus = WeightBus(pack_tensors(src), "cuda")
    stop = threading.Event()
    errs = []

    def publisher():
        g = 0
        while not stop.is_set():
            g += 1
            for t in pack_tensors(src).values():
                t.fill_(float(g % 200))
            bus.publish(src)

    def puller():
        ver = 0
        try:
            for _ in range(300):
                ver = bus.pull_into(dst, ver)
                vals = {float(t.reshape(-1)[0]) for t in
                        pack_tensors(dst).values()}
                uniform = all(
                    bool((t == t.reshape(-1)[0]).all())
                    for t in pack_tensors(dst).values())
                if len(vals) > 1 or not uniform:
                    errs.append(f"torn snapshot: {vals}")
                    return
        except Exception as e:  # pragma: no cover
            errs.append(repr(e))

    pub = threading.Thread(target=publisher, daemon=True)
    pub.start()
    puller()
    stop.set()
    pub.join(10)
    assert not errs, errs[:3]
