"""Full-shape soak of the host-replay transport path: a REAL ReplayBuffer
process (ingest + assemble + priority threads + logger + periodic
snapshots) racing a block producer that laps the ring, with a consumer
process that mimics the learner's staging depth and checks every batch for
tearing when it is EVICTED from staging (the worst-case lifetime a batch
can have under the shared-memory obs-slot pool — see worker._shared_tensor
and the in-flight bound comment at worker.ReplayBuffer._obs_pool).

Usage: python tools/shm_transport_soak.py [minutes]
"""

import sys
import time
import threading

sys.path.insert(0, ".")

import numpy as np
import torch
import torch.multiprocessing as mp


def consumer(bq, pq, stop_ev, out):
    """Drain batches holding a learner-like staging window; verify a batch
    only when it leaves the window (oldest possible read of its slot)."""
    from collections import deque
    torch.set_num_threads(1)
    staged = deque()
    n = 0
    errs = 0
    t0 = time.time()
    while not stop_ev.is_set():
        try:
            b = bq.get(timeout=1.0)
        except Exception:
            continue
        staged.append(b)
        if n and n % 500 == 0:
            print(f"[consumer] {n} batches, {errs} errors, "
                  f"{n / (time.time() - t0):.1f}/s", flush=True)
        if len(staged) > 5:
            old = staged.popleft()
            # tearing check: every row of a constant-filled block is uniform
            obs = old.obs.numpy()
            for i in range(obs.shape[0]):
                L = int(old.burn_in_steps[i] + old.learning_steps[i]
                        + old.forward_steps[i])
                row = obs[i, :L]
                if not (row == row.flat[0]).all():
                    errs += 1
                if not (obs[i, L:] == 0).all():
                    errs += 1
            pq.put((old.idxes, np.abs(np.random.randn(len(old.idxes)))
                    .astype(np.float32) + 0.5, old.old_ptr, 0.1,
                    old.old_count))
        n += 1
    out.put((n, errs, time.time() - t0))


def main(minutes=25.0):
    from r2d2_amd import config as cfg
    from r2d2_amd.train import _run_buffer
    from r2d2_amd.worker import ReplayBuffer, LocalBuffer

    c = cfg.apply("mspacman", gpu_replay=False, device="cpu",
                  buffer_capacity=24_000,        # 60 blocks: fast ring laps
                  learning_starts=4_000, training_steps=10**9,
                  log_interval=30, save_interval=10**9,
                  batch_queue_size=8, assemble_threads=2,
                  replay_snapshot_path="/tmp/soak_replay.snap",
                  replay_snapshot_interval=20.0)
    torch.set_num_threads(1)
    ctx = mp.get_context("fork")
    sq = [ctx.Queue()]
    bq, pq = ctx.Queue(8), ctx.Queue(8)
    buf = ReplayBuffer(sq, bq, pq)
    bp = ctx.Process(target=_run_buffer, args=(buf,), daemon=True)
    bp.start()

    def make_block(v):
        lb = LocalBuffer(c.action_dim)
        lb.reset(np.full(tuple(c.obs_shape), v, dtype=np.uint8))
        for t in range(c.block_length):
            lb.add(t % c.action_dim, 0.1,
                   np.full(tuple(c.obs_shape), v, dtype=np.uint8),
                   np.ones(c.action_dim, dtype=np.float32),
                   np.zeros((2, c.hidden_dim), dtype=np.float32))
        b, p, _ = lb.finish(np.zeros(c.action_dim, dtype=np.float32))
        return b, p

    pool = [make_block(v) for v in range(1, 17)]
    stop = threading.Event()
    sent = [0]

    def producer():
        rng = np.random.default_rng(0)
        while not stop.is_set():
            b, p = pool[sent[0] % len(pool)]
            sq[0].put((b, p + rng.random(len(p)).astype(np.float32), None))
            sent[0] += 1
            time.sleep(0.05)    # ~20 blocks/s -> a full ring lap every 3 s

    prod = threading.Thread(target=producer, daemon=True)
    prod.start()

    stop_ev = ctx.Event()
    out = ctx.Queue()
    cons = ctx.Process(target=consumer, args=(bq, pq, stop_ev, out),
                       daemon=True)
    cons.start()

    deadline = time.time() + minutes * 60
    while time.time() < deadline:
        time.sleep(10)
        if not bp.is_alive():
            raise SystemExit("FAIL: buffer process died")
        if not cons.is_alive():
            raise SystemExit("FAIL: consumer process died")
    stop.set()
    stop_ev.set()
    n, errs, el = out.get(timeout=30)
    print(f"SOAK RESULT: {n} batches in {el / 60:.1f} min "
          f"({n / el:.1f}/s), {sent[0]} blocks ingested "
          f"(~{sent[0] / 60:.0f} ring laps), torn/pad errors: {errs}")
    assert errs == 0, "tearing detected"
    print("SOAK OK")
    bp.terminate()
    cons.terminate()


if __name__ == "__main__":
    main(float(sys.argv[1]) if len(sys.argv) > 1 else 25.0)
