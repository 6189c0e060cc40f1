"""Multi-process (gloo, world_size=2) tests of the bucketed grad all-reducer.
Runs on CPU here; the same code path runs RCCL on MI355X."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from r2d2_amd.parallel.ddp import GradAllReducer


def _worker(rank, world_size, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        torch.manual_seed(0)  # same init on both ranks
        model = torch.nn.Sequential(
            torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
        reducer = GradAllReducer(list(model.parameters()), bucket_bytes=1024)

        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(8, 16)
        y = torch.randn(8, 4)

        reducer.prepare()
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        reducer.finish()
        grads = [p.grad.clone() for p in model.parameters()]

        # compute the expected average-of-ranks gradient locally
        dist.barrier()
        expect = []
        for p in model.parameters():
            p.grad = None
        model2 = model
        total = torch.zeros(1)
        per_rank_grads = []
        for r in range(world_size):
            torch.manual_seed(100 + r)
            xr = torch.randn(8, 16)
            yr = torch.randn(8, 4)
            for p in model2.parameters():
                p.grad = None
            reducer.detach()
            ((model2(xr) - yr) ** 2).mean().backward()
            per_rank_grads.append([p.grad.clone() for p in model2.parameters()])
        for tensors in zip(*per_rank_grads):
            expect.append(sum(tensors) / world_size)
        for g, e in zip(grads, expect):
            assert torch.allclose(g, e, atol=1e-6), (g - e).abs().max()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def _free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(300)
def test_torchrun_multirank_bench_path():
    """The EXACT multi-rank entry the driver uses — torch.distributed.run
    with --nproc-per-node 2 on 127.0.0.1 — must run bench.py end-to-end
    (rendezvous, per-rank Learner, max-over-ranks timing, rank-0 JSON).
    Uses the cartpole preset so the CPU step is fast; the launch/rendezvous/
    reduce code path is identical to the GPU run."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batches", "1",
         "--preset", "cartpole"],
        cwd=repo, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert line, out.stdout
    rec = json.loads(line[-1])
    assert rec["n_gpus"] == 2 and rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0


@pytest.mark.timeout(120)
def test_bucketed_allreduce_matches_mean_grad():
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, fail_q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(100)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs
    assert all(p.exitcode == 0 for p in procs)
