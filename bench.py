#!/usr/bin/env python3
"""Flagship benchmark: learner seq-samples/sec (bs=64, L=80, 84x84x4 frames).

Measures the full R2D2 learner update — target-net forward, online forward,
double-Q target with value rescaling, loss, backward, grad-norm clip, Adam,
and per-sequence priority computation — on synthetic 84x84x4 uint8 frames
with random-init weights (no datasets/checkpoints are downloadable in this
environment).  This is BASELINE.json's metric on BASELINE.json's config;
value is the WHOLE-JOB aggregate over all ranks.

Single GPU:     python bench.py --gpus 1 --steps 30 --warmup 10
Multi-GPU:      torchrun --nproc-per-node N bench.py --gpus N ...
                (one rank per GPU over RCCL; reads RANK/WORLD_SIZE/MASTER_*)
"""

import argparse
import json
import os
import time

import numpy as np
import torch


def build_batch(c, device, seed, full_len=True, hwc=False):
    """One synthetic device-resident training batch of the benchmark shape:
    B sequences of burn_in+learning+forward steps of (C,84,84) uint8 frames.
    ``hwc`` stores frames channels-innermost — the layout the GPU-resident
    replay keeps and the NHWC conv kernels consume directly."""
    from r2d2_amd.worker import TrainingBatch

    g = torch.Generator(device="cpu").manual_seed(seed)
    B = c.batch_size
    burn = torch.full((B,), c.burn_in_steps, dtype=torch.int64)
    learn = torch.full((B,), c.learning_steps, dtype=torch.int64)
    fwd = torch.full((B,), c.forward_steps, dtype=torch.int64)
    T = int((burn + learn + fwd).max())
    A = c.action_dim
    sum_learn = int(learn.sum())

    shape = (tuple(c.obs_shape[1:]) + (c.obs_shape[0],)) if hwc \
        else tuple(c.obs_shape)
    obs = torch.randint(0, 256, (B, T) + shape,
                        dtype=torch.uint8, generator=g)
    la = torch.zeros(B, T, A)
    la[torch.arange(B)[:, None], torch.arange(T)[None, :],
       torch.randint(0, A, (B, T), generator=g)] = 1.0
    lr = torch.randn(B, T, generator=g) * 0.1
    hidden = torch.randn(2, B, c.hidden_dim, generator=g) * 0.05
    action = torch.randint(0, A, (sum_learn, 1), generator=g)
    n_step_reward = torch.randn(sum_learn, generator=g).abs()
    gamma = torch.full((sum_learn,), c.gamma ** c.forward_steps)
    is_weights = torch.rand(sum_learn, generator=g) * 0.5 + 0.5

    batch = TrainingBatch(
        obs=obs, last_action=la, last_reward=lr, hidden=hidden,
        action=action, n_step_reward=n_step_reward, gamma=gamma,
        burn_in_steps=burn, learning_steps=learn, forward_steps=fwd,
        idxes=np.arange(B), is_weights=is_weights, old_ptr=0, env_steps=0)
    return batch.to(device, non_blocking=False)


def build_synthetic_block(c, rng):
    """A full actor Block of synthetic frames (for pre-filling the replay)."""
    from r2d2_amd.worker import Block

    steps, burn, learn, n = (c.block_length, c.burn_in_steps,
                             c.learning_steps, c.forward_steps)
    nseq = steps // learn
    rows = burn + steps + 1
    A = c.action_dim
    obs = rng.integers(0, 256, size=(rows,) + tuple(c.obs_shape), dtype=np.uint8)
    la = np.zeros((rows, A), dtype=bool)
    la[np.arange(rows), rng.integers(0, A, rows)] = True
    lr = rng.normal(size=rows).astype(np.float32) * 0.1
    action = rng.integers(0, A, steps).astype(np.uint8)
    nsr = rng.normal(size=steps).astype(np.float32).clip(-5, 5)
    gamma = np.full(steps, c.gamma ** n, dtype=np.float32)
    hidden = (rng.normal(size=(nseq, 2, c.hidden_dim)) * 0.05).astype(np.float32)
    learn_arr = np.full(nseq, learn, dtype=np.uint8)
    # LocalBuffer invariant: forward steps are bounded by the block end
    # (worker.py finish(): forward[-1] == 1) — the obs rows only extend one
    # step past the last learning step.
    fwd_arr = np.minimum(
        n, steps + 1 - np.cumsum(learn_arr.astype(np.int64))).astype(np.uint8)
    return Block(obs, la, lr, action, nsr, gamma, hidden, nseq,
                 np.full(nseq, burn, dtype=np.uint8), learn_arr, fwd_arr)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--preset", type=str, default="mspacman")
    ap.add_argument("--engine", type=str, default="auto",
                    choices=["auto", "hip", "eager"],
                    help="auto: HIP kernels on GPU when built, eager otherwise")
    ap.add_argument("--batches", type=int, default=4,
                    help="distinct synthetic batches rotated through")
    ap.add_argument("--gpu-replay", dest="gpu_replay", action="store_true",
                    default=None,
                    help="sample each step from the GPU-resident prioritized "
                         "replay (sum-tree + on-device gather in the timed "
                         "loop).  DEFAULT on GPU with the HIP engine — the "
                         "headline number times the full learner loop "
                         "(sample + gather + train + priority update)")
    ap.add_argument("--no-gpu-replay", dest="gpu_replay", action="store_false",
                    help="time prebuilt device batches instead")
    ap.add_argument("--replay-transitions", type=int, default=160_000,
                    help="synthetic transitions pre-filled into the GPU replay")
    args = ap.parse_args()

    from r2d2_amd import config as cfg

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    have_cuda = torch.cuda.is_available()
    # ranks beyond the visible GPU count share GPU 0 — lets the exact
    # multi-rank RCCL code path (init, reduce, max-over-ranks) be validated
    # on a 1-GPU box before an 8-GPU node is available
    dev_idx = local_rank % max(1, torch.cuda.device_count()) if have_cuda else 0
    device = torch.device(f"cuda:{dev_idx}" if have_cuda else "cpu")
    if have_cuda:
        torch.cuda.set_device(device)

    if world_size > 1:
        import torch.distributed as dist
        # R2D2_DIST_BACKEND=gloo lets the multi-rank GPU code path run with
        # several ranks SHARING one GPU (RCCL refuses duplicate devices);
        # the real 8-GPU run uses RCCL ("nccl" on ROCm) unchanged.
        backend = os.environ.get("R2D2_DIST_BACKEND") or (
            "nccl" if have_cuda else "gloo")
        dist.init_process_group(backend, rank=rank, world_size=world_size)

    c = cfg.apply(args.preset, device="cuda" if have_cuda else "cpu")
    use_hip = args.engine == "hip" or (args.engine == "auto" and have_cuda)
    if not have_cuda:
        c = cfg.apply(args.preset, device="cpu", dtype="fp32", amp=False,
                      use_hip_kernels=False)

    torch.manual_seed(1234)  # identical init on all ranks before broadcast
    from r2d2_amd.models.network import Network
    from r2d2_amd.worker import Learner

    model = Network(c.action_dim, c.obs_shape, c.hidden_dim, encoder=c.encoder,
                    forward_steps=c.forward_steps, mlp_hidden=c.mlp_hidden)
    learner = Learner(batch_queue=None, priority_queue=None, model=model)
    if use_hip and c.use_hip_kernels:
        try:
            learner.enable_hip_engine()
        except AttributeError:
            pass  # engine not built yet; eager path
    replay = None
    use_replay = args.gpu_replay
    if use_replay is None:   # default: full loop when the HIP engine runs
        use_replay = bool(have_cuda and learner.hip_engine
                          and len(c.obs_shape) == 3)
    if use_replay and have_cuda:
        assert learner.hip_engine, "--gpu-replay needs the HIP engine"
        from r2d2_amd.replay.gpu_replay import GpuReplayBuffer
        replay = GpuReplayBuffer(device=device,
                                 capacity=args.replay_transitions)
        rng = np.random.default_rng(42 + rank)
        # prefill (OUTSIDE the timed region): rotate a small pool of
        # prebuilt random blocks with fresh per-ingest priorities — frame
        # CONTENT doesn't affect learner throughput, and building 400
        # distinct 12 MB random blocks costs ~10 s of host RNG per rank
        # before the clock starts (noticeable at N=8)
        pool = [build_synthetic_block(c, rng) for _ in range(8)]
        for i in range(replay.num_blocks):
            replay.ingest(pool[i % len(pool)],
                          rng.random(replay.spb).astype(np.float32) + 0.1)
        torch.cuda.synchronize()
        batches = None
    else:
        # HWC frames only when the HIP engine is the consumer (the eager
        # Network expects NCHW; the replay stores HWC natively)
        hwc = bool(getattr(learner, "engine", None) is not None
                   and len(c.obs_shape) == 3)
        batches = [build_batch(c, device, seed=1000 + rank * 100 + i, hwc=hwc)
                   for i in range(args.batches)]

    pending = [None]

    def one_step(i):
        if replay is not None:
            # prefetch: the NEXT batch samples/gathers on the replay's side
            # stream while this step trains (one step of priority staleness,
            # far inside the reference's own <=12-batch staleness)
            batch = pending[0] if pending[0] is not None else replay.sample()
            tok = replay.sample_async()
            loss, prio = learner.train_step(batch)
            replay.update_priorities(batch.idxes, prio, batch.old_ptr)
            pending[0] = replay.sample_wait(tok)
        else:
            loss, _ = learner.train_step(batches[i % len(batches)])
        return loss

    def barrier():
        if world_size > 1:
            import torch.distributed as dist
            dist.barrier()
        if have_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        one_step(i)
    barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(args.warmup + i)
    if have_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    barrier()

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if world_size > 1 and have_cuda else "cpu")
    if world_size > 1:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    if (os.environ.get("R2D2_ENGINE_TIMING")
            and getattr(learner, "engine", None) and rank == 0):
        import sys
        agg = {}
        for name, ms in learner.engine.timing_report():
            agg[name] = agg.get(name, 0.0) + ms
        for name, ms in sorted(agg.items(), key=lambda kv: -kv[1]):
            print(f"  [stage] {name:12s} "
                  f"{ms / (args.warmup + args.steps):8.3f} ms/step",
                  file=sys.stderr)

    ms_per_step = elapsed / args.steps * 1000.0
    n_gpus = world_size if world_size > 1 else args.gpus
    value = c.batch_size * args.steps * n_gpus / elapsed
    baseline = 363.0  # BASELINE.md derived learner seq-samples/sec

    # BASELINE.json's headline metric name belongs to the default preset;
    # alternate presets get an honest label
    metric = "learner seq-samples/sec (bs=64, L=80, 84x84x4 frames)"
    if args.preset != "mspacman":
        metric = f"learner seq-samples/sec ({args.preset} preset)"
        baseline = None   # BASELINE.md's 363 belongs to the headline config

    if rank == 0:
        print(json.dumps({
            "metric": metric,
            "value": round(value, 2),
            "unit": "seq/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (round(value / baseline, 3)
                            if baseline else None),
            "dtype": "bf16" if (have_cuda and c.dtype == "bf16") else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"r2d2-{c.encoder}-cnn-lstm{c.hidden_dim}-dueling",
                "global_batch": c.batch_size * n_gpus,
                "seq_len": c.burn_in_steps + c.learning_steps,
                "burn_in": c.burn_in_steps,
                "forward_steps": c.forward_steps,
                "obs": list(c.obs_shape),
                "parallelism": f"dp{n_gpus}",
                "engine": "hip" if (use_hip and getattr(learner, "hip_engine", None)) else "eager",
                "replay": "gpu-resident" if replay is not None else "prebuilt-batches",
            },
        }))

    if world_size > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
