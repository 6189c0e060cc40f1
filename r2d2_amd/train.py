"""Process orchestration: N actor processes + 1 replay-buffer process +
learner in the main process (reference: /root/reference/train.py:20-44).

The epsilon ladder is the reference's: eps_i = base_eps^(1 + i/(N-1) * alpha)
(train.py:15-17).
"""

import random
import threading

import numpy as np
import torch
import torch.multiprocessing as mp

from . import config as cfg
from .models.network import Network
from .worker import Actor, Learner, ReplayBuffer


def epsilon_ladder(num_actors=None, base_eps=None, alpha=None):
    c = cfg.get()
    n = num_actors or c.num_actors
    base = base_eps or c.base_eps
    a = alpha or c.alpha_eps
    if n == 1:
        return [base]
    return [base ** (1 + i / (n - 1) * a) for i in range(n)]


def _run_actor(epsilon, model, sample_queue, seed, config_dict=None):
    if config_dict:   # spawn-context children start with default config
        cfg.apply(**config_dict)
    actor = Actor(epsilon, model, sample_queue, seed=seed)
    actor.run()


def _run_vector_actor(epsilons, model, sample_queues, device, seed,
                      config_dict=None, weight_bus=None):
    if config_dict:
        cfg.apply(**config_dict)
    from .worker import VectorActor
    va = VectorActor(epsilons, model, sample_queues, device=device, seed=seed,
                     weight_bus=weight_bus)
    va.run()


def _run_buffer(buffer: ReplayBuffer = None, queues=None, config_dict=None,
                restore_path=None, initial_training_steps=0):
    if buffer is None:
        # spawn context: a ReplayBuffer (threading.Lock inside) cannot be
        # pickled — build it in the child from the handed-over config
        cfg.apply(**config_dict)
        sq, bq, pq = queues
        buffer = ReplayBuffer(sq, bq, pq, restore_path=restore_path,
                              initial_training_steps=initial_training_steps)
    buffer.run()


def _latest_checkpoint(model_dir: str, game_name: str):
    """Newest ``{game}{N}.pth`` under model_dir, or None."""
    from .evaluate import _checkpoint_paths
    found = list(_checkpoint_paths(model_dir, game_name, 0))
    return found[-1][1] if found else None


def train(seed: int = 0, restart_dead_actors: bool = True,
          _force_spawn: bool = False, resume: str = None,
          model_dir: str = "models"):
    """Run the full training topology (reference train.py:20-44).

    ``resume``: a checkpoint path, or ``"auto"`` for the newest
    ``{game}{N}.pth`` in ``model_dir``.  Restores the learner (weights,
    update/env-step counters, and — via the ``.train.pth`` sidecar —
    optimizer moments and target net), and, when
    ``config.replay_snapshot_path`` points at an existing snapshot, the
    host replay contents too; the reference has no resume path at all
    (SURVEY §5)."""
    torch.manual_seed(seed)
    np.random.seed(seed)
    random.seed(seed)
    torch.set_num_threads(1)

    c = cfg.get()
    # CUDA is initialized in this process (the learner) before actors are
    # started: forked children cannot re-initialize CUDA, so a GPU
    # VectorActor child must be SPAWNED (with the live config handed over —
    # spawned interpreters start from the default preset).
    use_spawn = (c.vector_actors and c.actor_device == "cuda") or _force_spawn
    ctx = mp.get_context("spawn" if use_spawn else "fork")
    from dataclasses import asdict
    config_dict = asdict(c) if use_spawn else None

    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder=c.encoder,
                    forward_steps=c.forward_steps, mlp_hidden=c.mlp_hidden)
    model.share_memory()

    sample_queues = [ctx.Queue() for _ in range(c.num_actors)]
    batch_queue = ctx.Queue(c.batch_queue_size)
    priority_queue = ctx.Queue(c.batch_queue_size)

    use_gpu_replay = c.gpu_replay and torch.cuda.is_available()
    learner = Learner(batch_queue, priority_queue, model,
                      model_dir=model_dir)
    if use_gpu_replay and c.use_hip_kernels:
        learner.enable_hip_engine()
    if resume:
        path = (_latest_checkpoint(model_dir, c.game_name)
                if resume == "auto" else resume)
        if path is None:
            print(f"[train] resume='auto': no checkpoint under {model_dir}/; "
                  f"starting fresh")
        else:
            print(f"[train] resuming from {path}")
            learner.load_checkpoint(path)
    import os
    snap = c.replay_snapshot_path
    restore = snap if (resume and snap and os.path.exists(snap)) else None
    if restore:
        print(f"[train] restoring replay snapshot {restore}")
    # spawn mode rebuilds the buffer in the child (_run_buffer), which does
    # the restore there — don't load the snapshot twice
    buffer = None if use_gpu_replay else ReplayBuffer(
        sample_queues, batch_queue, priority_queue,
        restore_path=None if use_spawn else restore,
        initial_training_steps=learner.num_updates)

    if c.vector_actors:
        # one driver process, all envs in lockstep, batched inference.
        # The learner's WeightBus (device-resident prepacked weights +
        # version word) travels to the spawned child via CUDA IPC — pulls
        # become same-GPU slice copies instead of CPU state_dict loads.
        spawners = [lambda: ctx.Process(
            target=_run_vector_actor,
            args=(epsilon_ladder(), model, sample_queues, c.actor_device,
                  seed + 1, config_dict, learner.weight_bus))]
    else:
        spawners = [
            (lambda eps=eps, i=i: ctx.Process(
                target=_run_actor,
                args=(eps, model, sample_queues[i], seed + 1 + i,
                      config_dict)))
            for i, eps in enumerate(epsilon_ladder())]
    actor_procs = [s() for s in spawners]
    for p in actor_procs:
        p.start()

    buffer_proc = None
    if buffer is not None:
        if use_spawn:
            buffer_proc = ctx.Process(target=_run_buffer, kwargs=dict(
                queues=(sample_queues, batch_queue, priority_queue),
                config_dict=config_dict, restore_path=restore,
                initial_training_steps=learner.num_updates))
        else:
            buffer_proc = ctx.Process(target=_run_buffer, args=(buffer,))
        buffer_proc.start()

    # actor supervision: the reference silently loses dead actor processes
    # (SURVEY §5 — throughput degrades with no signal); restart them.
    stop = threading.Event()

    restarts = [0] * len(actor_procs)
    max_restarts = 10

    def _watchdog():
        while not stop.wait(5.0):
            for i, p in enumerate(actor_procs):
                if not p.is_alive() and restarts[i] <= max_restarts:
                    restarts[i] += 1
                    if restarts[i] > max_restarts:
                        # persistent crash (bad env/config): stop the
                        # restart storm and surface it instead
                        print(f"[train] actor {i} died {max_restarts} "
                              f"times; giving up on it")
                        continue
                    print(f"[train] actor process {i} died "
                          f"(exitcode {p.exitcode}); restarting "
                          f"({restarts[i]}/{max_restarts})")
                    actor_procs[i] = spawners[i]()
                    actor_procs[i].start()

    wd = None
    if restart_dead_actors:
        wd = threading.Thread(target=_watchdog, daemon=True)
        wd.start()

    if use_gpu_replay:
        # configs[2]/[3]: the learner owns the GPU-resident replay; no
        # host buffer process
        learner.run_with_gpu_replay(sample_queues)
    else:
        learner.run()

    stop.set()
    if buffer_proc is not None:
        buffer_proc.join()
    for p in actor_procs:
        p.terminate()


def main():
    import argparse
    ap = argparse.ArgumentParser(description="R2D2 training (reference "
                                 "train.py topology)")
    ap.add_argument("--preset", type=str, default=None,
                    help="config preset (see r2d2_amd.config.PRESETS)")
    ap.add_argument("--resume", type=str, default=None,
                    help="checkpoint path, or 'auto' = newest in --model-dir")
    ap.add_argument("--model-dir", type=str, default="models")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    if args.preset:
        cfg.apply(args.preset)
    train(seed=args.seed, resume=args.resume, model_dir=args.model_dir)


if __name__ == "__main__":
    main()
