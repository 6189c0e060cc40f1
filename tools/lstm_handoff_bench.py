"""Microbench: price the LSTM producer-flag handoff protocol per step at a
given workgroup count (the per-step latency floor of the persistent LSTM
kernels — ops/hip/lstm_kernels.hip)."""

import sys
import time

import torch

sys.path.insert(0, ".")
from r2d2_amd.ops import hip_ops  # noqa: E402

m = hip_ops.ext(required=True)
steps = 85
for kind, fn in (("flags", m.handoff_bench),
                 ("counter", m.handoff_counter_bench)):
    for nblocks in (32, 64, 128, 256):
        bar = torch.zeros(512, dtype=torch.int32, device="cuda")
        fn(bar, steps, nblocks)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 20
        for _ in range(reps):
            fn(bar, steps, nblocks)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / reps / steps * 1e6
        print(f"{kind:8s} nblocks={nblocks:4d}: {us:7.2f} us/step "
              f"({us * steps * 1e-3:6.2f} ms per {steps}-step pass)")
