"""Bucketed data-parallel gradient all-reduce over RCCL/xGMI.

One process per GPU (`torch.distributed`, backend "nccl" == RCCL on ROCm).
Gradients are packed into flat buckets in reverse parameter order (the order
backward produces them) and each bucket's all-reduce is launched as soon as
its last gradient arrives, overlapping communication with the rest of
backward.  xGMI is point-to-point (7 links x ~153 GB/s per GPU), so ring
all-reduce is per-link bound: bucket sizes default to 8 MiB to keep several
reduces in flight across links rather than one big serial ring.

The reference has no multi-learner path at all; this module
is the MI355X-native replacement for its single `cuda` learner.
"""

from typing import List, Optional

import torch
import torch.distributed as dist


class GradAllReducer:
    """Attach to a model; call ``prepare()`` before each backward and
    ``finish()`` after; gradients end up averaged across ranks."""

    def __init__(self, params: List[torch.nn.Parameter],
                 bucket_bytes: int = 8 << 20,
                 process_group: Optional[object] = None,
                 async_op: bool = True):
        self.params = [p for p in params if p.requires_grad]
        self.group = process_group
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.async_op = async_op and self.world_size > 1
        self.enabled = self.world_size > 1

        # buckets in reverse order (backward completes roughly in this order)
        self.buckets: List[List[torch.nn.Parameter]] = []
        cur, cur_bytes = [], 0
        for p in reversed(self.params):
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= bucket_bytes:
                self.buckets.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(cur)

        self._bucket_of = {}
        for bi, bucket in enumerate(self.buckets):
            for p in bucket:
                self._bucket_of[p] = bi

        self._flat = [None] * len(self.buckets)
        self._pending = [0] * len(self.buckets)
        self._works = [None] * len(self.buckets)
        self._hooks = []
        if self.enabled:
            for p in self.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad))

    # ------------------------------------------------------------------
    def _on_grad(self, p: torch.nn.Parameter):
        bi = self._bucket_of[p]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            self._launch(bi)

    def _launch(self, bi: int):
        bucket = self.buckets[bi]
        flat = torch._utils._flatten_dense_tensors([p.grad for p in bucket])
        flat.div_(self.world_size)
        work = dist.all_reduce(flat, group=self.group, async_op=self.async_op)
        self._flat[bi] = flat
        self._works[bi] = work

    # ------------------------------------------------------------------
    def prepare(self):
        if not self.enabled:
            return
        for bi, bucket in enumerate(self.buckets):
            self._pending[bi] = len(bucket)
            self._works[bi] = None
            self._flat[bi] = None

    def finish(self):
        if not self.enabled:
            return
        for bi, bucket in enumerate(self.buckets):
            if self._pending[bi] > 0 and self._flat[bi] is None:
                # params that never got grads this step (unused): treat their
                # grads as zero and reduce what exists
                for p in bucket:
                    if p.grad is None:
                        p.grad = torch.zeros_like(p)
                self._launch(bi)
            work = self._works[bi]
            if work is not None and self.async_op:
                work.wait()
            flat = self._flat[bi]
            if flat is not None:
                outs = torch._utils._unflatten_dense_tensors(
                    flat, [p.grad for p in bucket])
                for p, g in zip(bucket, outs):
                    p.grad.copy_(g)

    def reduce_all(self):
        """Synchronous bucket all-reduce of existing .grads (for the manual-
        backward HIP engine path, where no autograd hooks fire)."""
        if not self.enabled:
            return
        works = []
        flats = []
        for bucket in self.buckets:
            for p in bucket:
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
            flat = torch._utils._flatten_dense_tensors([p.grad for p in bucket])
            flat.div_(self.world_size)
            works.append(dist.all_reduce(flat, group=self.group, async_op=True))
            flats.append(flat)
        for bucket, work, flat in zip(self.buckets, works, flats):
            work.wait()
            outs = torch._utils._unflatten_dense_tensors(
                flat, [p.grad for p in bucket])
            for p, g in zip(bucket, outs):
                p.grad.copy_(g)

    def broadcast_params(self, src: int = 0):
        if not self.enabled:
            return
        with torch.no_grad():
            for p in self.params:
                dist.broadcast(p.data, src=src, group=self.group)

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
