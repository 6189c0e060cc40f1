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


@pytest.mark.timeout(300)
def test_train_spawn_context(tmp_path, monkeypatch):
    """The spawn-context actor path (used when vector actors run on cuda):
    spawned children must receive the live config and shared model."""
    monkeypatch.chdir(tmp_path)
    tiny_cfg(tmp_path, vector_actors=True)
    from r2d2_amd.train import train
    train(seed=0, _force_spawn=True)
    assert (tmp_path / "metrics.jsonl").exists()
