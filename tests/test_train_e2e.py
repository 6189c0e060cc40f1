"""Full multi-process train() orchestration on the CartPole CPU config:
actor processes (or the VectorActor driver) -> replay-buffer process ->
learner in the main process, through real mp.Queues, to completion
(reference train.py:20-44 topology)."""

import os

import pytest
import torch.multiprocessing as mp

from r2d2_amd import config as cfg


def tiny_cfg(tmp_path, **kw):
    base = dict(buffer_capacity=1600, block_length=16, burn_in_steps=4,
                learning_steps=4, forward_steps=2, batch_size=8,
                learning_starts=200, hidden_dim=32, mlp_hidden=32,
                training_steps=30, num_actors=2, max_episode_steps=60,
                log_interval=1, save_interval=10_000,
                metrics_path=str(tmp_path / "metrics.jsonl"))
    base.update(kw)
    return cfg.apply("cartpole", **base)


@pytest.mark.timeout(300)
@pytest.mark.parametrize("vector", [False, True])
def test_train_to_completion(tmp_path, vector, monkeypatch):
    monkeypatch.chdir(tmp_path)
    tiny_cfg(tmp_path, vector_actors=vector)
    from r2d2_amd.train import train
    train(seed=0)
    # learner finished all updates; metrics JSONL was emitted
    assert (tmp_path / "metrics.jsonl").exists()


@pytest.mark.timeout(600)
def test_train_resume_continues(tmp_path, monkeypatch):
    """Elastic resume through the train() entry: a finished run's newest
    checkpoint (resume='auto') restores the learner counters; the replay
    snapshot (config.replay_snapshot_path) restores the buffer contents,
    and the buffer's termination condition counts from the restored update
    number (a resumed learner doing N more updates must not leave the
    buffer process waiting for 2N)."""
    monkeypatch.chdir(tmp_path)
    snap = str(tmp_path / "replay.snap")
    tiny_cfg(tmp_path, training_steps=15, save_interval=15,
             replay_snapshot_path=snap, replay_snapshot_interval=0.0)
    from r2d2_amd.train import train
    train(seed=0)
    assert (tmp_path / "models" / "CartPole15.pth").exists()

    # second leg: 15 more updates on top of the restored 15
    tiny_cfg(tmp_path, training_steps=30, save_interval=15,
             replay_snapshot_path=snap, replay_snapshot_interval=0.0)
    train(seed=1, resume="auto")
    assert (tmp_path / "models" / "CartPole30.pth").exists()
    import torch
    _, num_updates, env_steps, _ = torch.load(
        str(tmp_path / "models" / "CartPole30.pth"), map_location="cpu",
        weights_only=False)
    assert num_updates == 30 and env_steps > 0


@pytest.mark.timeout(300)
def test_train_spawn_context(tmp_path, monkeypatch):
    """The spawn-context actor path (used when vector actors run on cuda):
    spawned children must receive the live config and shared model."""
    monkeypatch.chdir(tmp_path)
    tiny_cfg(tmp_path, vector_actors=True)
    from r2d2_amd.train import train
    train(seed=0, _force_spawn=True)
    assert (tmp_path / "metrics.jsonl").exists()
