"""Evaluation harness + checkpoint/resume tests (reference test.py:14-88 and
the 4-tuple checkpoint contract at reference worker.py:380-381)."""

import os
import queue

import numpy as np
import pytest
import torch

from r2d2_amd import config as cfg
from r2d2_amd import evaluate
from r2d2_amd.models.network import Network
from r2d2_amd.worker import Learner


def setup_cartpole(**kw):
    base = dict(buffer_capacity=640, block_length=40, burn_in_steps=8,
                learning_steps=8, forward_steps=3, batch_size=8,
                learning_starts=80, hidden_dim=32, mlp_hidden=32,
                training_steps=5, num_actors=1, max_episode_steps=50,
                save_interval=2)
    base.update(kw)
    return cfg.apply("cartpole", **base)


def make_learner(tmp_path, seed=0):
    c = setup_cartpole()
    torch.manual_seed(seed)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="mlp",
                    forward_steps=c.forward_steps, mlp_hidden=c.mlp_hidden)
    learner = Learner(queue.Queue(), queue.Queue(), model,
                      model_dir=str(tmp_path))
    return c, learner


def test_checkpoint_4tuple_and_sidecar(tmp_path):
    c, learner = make_learner(tmp_path)
    learner.num_updates = 2
    learner.env_steps = 123
    import time
    learner.save(time.time() - 90)
    path = tmp_path / f"{c.game_name}2.pth"
    assert path.exists()
    state, n, steps, minutes = torch.load(path, weights_only=False)
    assert n == 2 and steps == 123
    assert 1.0 <= minutes <= 2.0
    # loadable by a fresh Network (the reference test.py:27,30 contract)
    net = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder="mlp",
                  forward_steps=c.forward_steps, mlp_hidden=c.mlp_hidden)
    net.load_state_dict(state)
    assert (tmp_path / f"{c.game_name}2.train.pth").exists()


def test_resume_roundtrip(tmp_path):
    from bench import build_batch
    c, learner = make_learner(tmp_path)
    dev = torch.device("cpu")
    batch = build_batch(c, dev, seed=3)
    for _ in range(3):
        learner.train_step(batch)
    learner.num_updates = 2
    learner.env_steps = 50
    import time
    learner.save(time.time())

    # fresh learner resumes: same params, same optimizer moments
    _, learner2 = make_learner(tmp_path, seed=99)
    learner2.load_checkpoint(str(tmp_path / f"{c.game_name}2.pth"))
    assert learner2.num_updates == 2 and learner2.env_steps == 50
    for p1, p2 in zip(learner.online_net.parameters(),
                      learner2.online_net.parameters()):
        assert torch.equal(p1, p2)
    s1 = learner.optimizer.state_dict()["state"]
    s2 = learner2.optimizer.state_dict()["state"]
    assert set(s1) == set(s2)
    for k in s1:
        assert torch.allclose(s1[k]["exp_avg"], s2[k]["exp_avg"])
    # training continues identically from the restored state
    l1, _ = learner.train_step(batch)
    l2, _ = learner2.train_step(batch)
    assert torch.allclose(torch.as_tensor(l1), torch.as_tensor(l2))


@pytest.mark.timeout(300)
def test_eval_harness(tmp_path):
    c, learner = make_learner(tmp_path)
    import time
    for n in (2, 4):
        learner.num_updates = n
        learner.env_steps = 100 * n
        learner.save(time.time() - 60)
    results = evaluate.test(model_dir=str(tmp_path), num_episodes=2,
                            pool_size=2, out_dir=str(tmp_path))
    assert len(results) == 2
    assert [r["num_updates"] for r in results] == [2, 4]
    for r in results:
        assert r["env_frames"] == r["env_steps"] * c.frame_skip
        assert len(r["rewards"]) == 2
        assert np.isfinite(r["mean_reward"])
    assert (tmp_path / f"{c.game_name}_eval.csv").exists()
    assert (tmp_path / f"{c.game_name}_eval.jsonl").exists()


def test_eval_harness_accepts_reference_checkpoints(tmp_path):
    """A checkpoint trained by the ORIGINAL reference (anonymous `feature`
    Sequential encoder keys) must evaluate through evaluate.test() via the
    key-mapping fallback — a reference user can point the harness at their
    existing models/ directory."""
    import time

    c = cfg.apply("mspacman", env_type="synthetic", obs_shape=(1, 84, 84),
                  action_dim=4, hidden_dim=32, max_episode_steps=8,
                  device="cpu", dtype="fp32", use_hip_kernels=False,
                  gpu_replay=False)
    torch.manual_seed(3)
    src = Network(4, obs_shape=(1, 84, 84), hidden_dim=32, encoder="nature")
    inv = {"encoder.conv1": "feature.0", "encoder.conv2": "feature.2",
           "encoder.conv3": "feature.4", "encoder.fc": "feature.7"}
    ref_sd = {}
    for k, v in src.state_dict().items():
        head, _, tail = k.rpartition(".")
        ref_sd[f"{inv.get(head, head)}.{tail}" if head else k] = v
    torch.save((ref_sd, 500, 2000, 1.5), tmp_path / f"{c.game_name}500.pth")

    results = evaluate.test(model_dir=str(tmp_path), num_episodes=1,
                            pool_size=1, out_dir=str(tmp_path))
    assert len(results) == 1 and results[0]["num_updates"] == 500
    assert np.isfinite(results[0]["mean_reward"])


def test_learner_resumes_from_reference_checkpoint(tmp_path):
    """Learner.load_checkpoint applies the same key-mapping fallback, so
    training can RESUME from a reference-trained checkpoint."""
    c = cfg.apply("mspacman", env_type="synthetic", obs_shape=(1, 84, 84),
                  action_dim=4, hidden_dim=32, device="cpu", dtype="fp32",
                  use_hip_kernels=False, gpu_replay=False, amp=False)
    torch.manual_seed(5)
    src = Network(4, obs_shape=(1, 84, 84), hidden_dim=32, encoder="nature")
    inv = {"encoder.conv1": "feature.0", "encoder.conv2": "feature.2",
           "encoder.conv3": "feature.4", "encoder.fc": "feature.7"}
    ref_sd = {}
    for k, v in src.state_dict().items():
        head, _, tail = k.rpartition(".")
        ref_sd[f"{inv.get(head, head)}.{tail}" if head else k] = v
    p = tmp_path / f"{c.game_name}700.pth"
    torch.save((ref_sd, 700, 4000, 2.0), p)

    model = Network(c.action_dim, c.obs_shape, c.hidden_dim,
                    encoder="nature")
    learner = Learner(queue.Queue(), queue.Queue(), model,
                      model_dir=str(tmp_path))
    learner.load_checkpoint(str(p))
    assert learner.num_updates == 700 and learner.env_steps == 4000
    assert torch.equal(learner.online_net.state_dict()["encoder.conv1.weight"],
                       src.state_dict()["encoder.conv1.weight"])
    # no sidecar: target net must equal the (remapped) online weights
    assert torch.equal(learner.target_net.state_dict()["encoder.conv1.weight"],
                       src.state_dict()["encoder.conv1.weight"])


def test_checkpoint_discovery_mixed_intervals(tmp_path):
    """Discovery must find every {game}{N}.pth regardless of the configured
    save_interval and survive gaps (the reference's k*interval walk stops
    at the first hole and finds NOTHING when the run used a different
    interval); .train.pth optimizer sidecars are not checkpoints."""
    from r2d2_amd.evaluate import _checkpoint_paths

    for n in (500, 2000, 10000):        # mixed intervals + a gap
        (tmp_path / f"Pong{n}.pth").touch()
        (tmp_path / f"Pong{n}.train.pth").touch()
    (tmp_path / "PongBest.pth").touch()  # non-numeric: ignored
    got = list(_checkpoint_paths(str(tmp_path), "Pong", save_interval=777))
    assert [n for n, _ in got] == [500, 2000, 10000]
    assert all(p.endswith(f"Pong{n}.pth") for n, p in got)
