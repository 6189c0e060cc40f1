"""GPU tests: device sum-tree vs CPU golden descent; GPU replay ingest/sample
round-trip; ring-stale priority masking."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from r2d2_amd.ops import hip_ops
    from r2d2_amd import config as cfg


def make_tree(num_seq, dev="cuda"):
    m = hip_ops.ext()
    levels = 1
    while (1 << levels) < num_seq:
        levels += 1
    leaf_offset = (1 << levels) - 1
    tree = torch.zeros(2 * (1 << levels) - 1, dtype=torch.float64, device=dev)
    return m, tree, leaf_offset, levels


def cpu_descent(tree_np, leaf_offset, levels, u):
    node = 0
    for _ in range(levels):
        left = 2 * node + 1
        if u < tree_np[left]:
            node = left
        else:
            u -= tree_np[left]
            node = left + 1
    return node - leaf_offset


def test_sumtree_update_and_sample_match_cpu():
    m, tree, leaf_offset, levels = make_tree(100)
    rng = np.random.default_rng(0)
    idx = torch.arange(100, dtype=torch.int64, device="cuda")
    td = torch.from_numpy(rng.random(100).astype(np.float32) * 5).cuda()
    m.sumtree_update(tree, leaf_offset, idx, td, 0.9, 0, 0, 10, 10)
    torch.cuda.synchronize()
    tree_np = tree.cpu().numpy()
    leaves = tree_np[leaf_offset:leaf_offset + 128]
    np.testing.assert_allclose(leaves[:100],
                               td.cpu().numpy().astype(np.float64) ** 0.9,
                               rtol=1e-6)
    assert abs(tree_np[0] - leaves.sum()) < 1e-9 * tree_np[0]
    # internal consistency at every node
    for node in range(leaf_offset):
        np.testing.assert_allclose(tree_np[node],
                                   tree_np[2 * node + 1] + tree_np[2 * node + 2],
                                   rtol=1e-9)

    # sampling with known jitter reproduces the CPU descent exactly
    jitter = torch.rand(64, device="cuda")
    out_idx, out_prio, out_w = m.sumtree_sample(tree, leaf_offset, levels,
                                                jitter, 64, 0.6,
                                                (1 << levels) - 1)
    torch.cuda.synchronize()
    total = tree_np[0]
    j = jitter.cpu().numpy().astype(np.float64)
    expect = [cpu_descent(tree_np, leaf_offset, levels,
                          (i + j[i]) * (total / 64)) for i in range(64)]
    np.testing.assert_array_equal(out_idx.cpu().numpy(), expect)
    # IS weights: (p/min)^'-beta
    p = out_prio.cpu().numpy().astype(np.float64)
    w_expect = (p / p[p > 0].min()) ** -0.6
    np.testing.assert_allclose(out_w.cpu().numpy(), w_expect, rtol=1e-5)


def test_sumtree_duplicate_updates_consistent():
    m, tree, leaf_offset, levels = make_tree(64)
    idx = torch.zeros(32, dtype=torch.int64, device="cuda")  # all same leaf
    td = torch.rand(32, device="cuda") + 0.5
    m.sumtree_update(tree, leaf_offset, idx, td, 1.0, 0, 0, 8, 8)
    torch.cuda.synchronize()
    tree_np = tree.cpu().numpy()
    # whatever write won at the leaf, the root must equal the leaf sum
    assert abs(tree_np[0] - tree_np[leaf_offset:].sum()) < 1e-9
    assert tree_np[leaf_offset] in td.cpu().numpy().astype(np.float64)


def test_sumtree_stale_mask():
    m, tree, leaf_offset, levels = make_tree(64)
    idx = torch.arange(64, dtype=torch.int64, device="cuda")
    m.sumtree_update(tree, leaf_offset, idx, torch.ones(64, device="cuda"),
                     1.0, 0, 0, 8, 8)
    # ring advanced old_ptr=0 -> cur_ptr=2: blocks 0,1 (leaves 0..15) stale
    upd_idx = torch.tensor([0, 8, 16, 40], dtype=torch.int64, device="cuda")
    m.sumtree_update(tree, leaf_offset, upd_idx,
                     torch.full((4,), 9.0, device="cuda"), 1.0, 0, 2, 8, 8)
    torch.cuda.synchronize()
    leaves = tree.cpu().numpy()[leaf_offset:leaf_offset + 64]
    assert leaves[0] == 1.0 and leaves[8] == 1.0      # masked (stale)
    assert leaves[16] == 9.0 and leaves[40] == 9.0    # applied
    # wraparound direction: old_ptr=6, cur_ptr=1 -> stale leaves >= 48 or < 8
    m.sumtree_update(tree, leaf_offset, upd_idx,
                     torch.full((4,), 5.0, device="cuda"), 1.0, 6, 1, 8, 8)
    torch.cuda.synchronize()
    leaves = tree.cpu().numpy()[leaf_offset:leaf_offset + 64]
    assert leaves[0] == 1.0                            # stale (< 8)
    assert leaves[8] == 5.0 and leaves[16] == 5.0 and leaves[40] == 5.0


def test_gpu_replay_roundtrip():
    c = cfg.apply("mspacman", buffer_capacity=8000, batch_size=16)
    from r2d2_amd.replay.gpu_replay import GpuReplayBuffer
    from bench import build_synthetic_block
    rng = np.random.default_rng(1)
    replay = GpuReplayBuffer(device="cuda", capacity=8000)
    blocks = []
    for v in range(replay.num_blocks):
        blk = build_synthetic_block(c, rng)
        blk.obs[:] = v + 1   # constant per block for content checks
        blocks.append(blk)
        replay.ingest(blk, np.ones(replay.spb, dtype=np.float32) * (v + 1))
    torch.cuda.synchronize()
    assert len(replay) == 8000
    # sum-tree invariant: root == sum of leaves after streamed ingest
    leaves = replay.tree[replay.leaf_offset:].sum()
    assert abs(float(replay.tree[0]) - float(leaves)) < 1e-6 * max(
        1.0, float(leaves))

    batch = replay.sample(16)
    torch.cuda.synchronize()
    B, T = 16, c.seq_len
    assert batch.obs.shape == (B, T, 84, 84, 4)  # HWC store
    assert batch.hidden.shape == (2, B, c.hidden_dim)
    R = int(batch.learning_steps.sum())
    assert batch.action.shape == (R, 1)
    assert batch.is_weights.shape == (R,)
    idx = batch.idxes.cpu().numpy()
    for i in range(B):
        bi = idx[i] // replay.spb
        si = idx[i] % replay.spb
        v = bi + 1
        L = int(batch.burn_in_steps[i] + batch.learning_steps[i]
                + batch.forward_steps[i])
        ob = batch.obs[i].cpu().numpy()
        assert (ob[:L] == v).all()
        assert (ob[L:] == 0).all()
        blk = blocks[bi]
        ls = int(blk.learning_steps[:si].sum())
        le = ls + int(batch.learning_steps[i])
        seg0 = int(batch.learning_steps[:i].sum())
        np.testing.assert_array_equal(
            batch.action[seg0:seg0 + le - ls, 0].cpu().numpy(),
            blk.action[ls:le].astype(np.int64))
        np.testing.assert_allclose(
            batch.n_step_reward[seg0:seg0 + le - ls].cpu().numpy(),
            blk.n_step_reward[ls:le], rtol=1e-6)
        np.testing.assert_allclose(batch.hidden[:, i].cpu().numpy(),
                                   blk.hidden[si], rtol=1e-6)
        # last_action one-hot and last_reward from the obs-aligned rows
        rows = slice(int(blk.burn_in_steps[0]) + ls - int(batch.burn_in_steps[i]),
                     int(blk.burn_in_steps[0]) + ls
                     + int(batch.learning_steps[i]) + int(batch.forward_steps[i]))
        la_idx = blk.last_action[rows].argmax(1)
        got_la = batch.last_action[i, :L].cpu().numpy().argmax(1)
        np.testing.assert_array_equal(got_la, la_idx)
        np.testing.assert_allclose(batch.last_reward[i, :L].cpu().numpy(),
                                   blk.last_reward[rows], rtol=1e-6)


def test_gpu_replay_train_integration():
    """Sample from GPU replay -> HIP train_step -> on-device priority update."""
    c = cfg.apply("mspacman", buffer_capacity=4000, batch_size=8)
    from r2d2_amd.replay.gpu_replay import GpuReplayBuffer
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner
    from bench import build_synthetic_block
    torch.manual_seed(0)
    rng = np.random.default_rng(2)
    replay = GpuReplayBuffer(device="cuda", capacity=4000)
    for _ in range(replay.num_blocks):
        replay.ingest(build_synthetic_block(c, rng),
                      rng.random(replay.spb).astype(np.float32) + 0.5)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    learner = Learner(None, None, model)
    learner.enable_hip_engine()
    total0 = replay.total_priority
    for _ in range(3):
        batch = replay.sample(8)
        loss, prio = learner.train_step(batch)
        assert torch.isfinite(loss)
        replay.update_priorities(batch.idxes, prio, batch.old_ptr)
    torch.cuda.synchronize()
    assert replay.total_priority != total0
    assert np.isfinite(replay.total_priority)


def test_device_positions_match_host_positions():
    """The on-device positions_meta path (used for every replay-sampled
    batch) must produce the same loss/priorities/grads as the host numpy
    positions path on the same batch."""
    c = cfg.apply("mspacman", buffer_capacity=4000, batch_size=8)
    from r2d2_amd.replay.gpu_replay import GpuReplayBuffer
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner
    from bench import build_synthetic_block
    torch.manual_seed(3)
    rng = np.random.default_rng(7)
    replay = GpuReplayBuffer(device="cuda", capacity=4000)
    for _ in range(replay.num_blocks):
        replay.ingest(build_synthetic_block(c, rng),
                      rng.random(replay.spb).astype(np.float32) + 0.5)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="nature",
                    forward_steps=c.forward_steps)
    learner = Learner(None, None, model)
    learner.enable_hip_engine()
    batch = replay.sample(8)
    assert batch.meta_dev is not None

    loss_d, prio_d = learner.engine.train_step(batch)
    grads_d = learner.engine.flat_grad.clone()

    batch.meta_dev = None   # force the host numpy positions path
    loss_h, prio_h = learner.engine.train_step(batch)
    grads_h = learner.engine.flat_grad.clone()

    assert torch.equal(loss_d, loss_h)
    assert torch.equal(prio_d, prio_h)
    # wgrad f32 atomics reorder between runs — grads match to ulp noise
    assert torch.allclose(grads_d, grads_h, rtol=1e-3, atol=1e-5), \
        float((grads_d - grads_h).abs().max())
