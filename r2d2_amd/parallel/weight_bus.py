"""Device-resident learner->actor weight distribution (the WeightBus).

The reference publishes weights by copying the learner's GPU state_dict to a
shared-memory CPU model every 4 updates, and each actor load_state_dict()s
from it (reference train.py:23, worker.py:306-307,560-566).  On MI355X the
VectorActor's inference runs on the SAME GPU as the learner, so the round
trip GPU -> CPU -> GPU (plus a full prepack refresh per pull) is pure waste.

The WeightBus is two flat device buffers (bf16 + f32) holding the learner's
PREPACKED inference weights plus an int32 version word:

- the learner publishes by device-to-device copying its engine pack tensors
  into the bus slices and bumping the version — a handful of D2D copies
  (~8.5 MB at ~8 TB/s HBM3E), launched on the learner's stream;
- the VectorActor process (which received the bus tensors through spawn
  pickling = CUDA IPC) polls the version word and, when it moved, copies
  the slices STRAIGHT into its HipInference pack tensors — no state_dict,
  no repack, no host round trip.

Torn reads are prevented by re-checking the version after the snapshot and
redoing it if a publish overlapped.
"""

from typing import Dict, List, Tuple

import torch


def _iter_tensors(obj, prefix: str):
    """Yield (path, tensor) for every cuda tensor reachable from a pack:
    direct tensor attributes, dict-of-tensor attributes, and one level of
    nested pack objects (e.g. the IMPALA sub-pack)."""
    for name in sorted(vars(obj)):
        if name.startswith("_"):
            continue
        v = vars(obj)[name]
        p = f"{prefix}{name}"
        if torch.is_tensor(v):
            if v.is_cuda and v.is_floating_point():
                yield p, v
        elif isinstance(v, dict):
            for k in sorted(v, key=repr):
                t = v[k]
                if torch.is_tensor(t) and t.is_cuda and t.is_floating_point():
                    yield f"{p}[{k!r}]", t
        elif hasattr(v, "__dict__") and hasattr(v, "refresh"):
            yield from _iter_tensors(v, p + ".")


def pack_tensors(pack) -> Dict[str, torch.Tensor]:
    """Ordered {path: tensor} of a _NetPack's floating device tensors."""
    return dict(_iter_tensors(pack, ""))


class WeightBus:
    """Flat bf16+f32 publish buffers over a pack tensor spec."""

    def __init__(self, spec: Dict[str, torch.Tensor], device):
        self.device = torch.device(device)
        self.layout: List[Tuple[str, torch.dtype, torch.Size, int]] = []
        sizes = {torch.bfloat16: 0, torch.float32: 0}
        for name, t in spec.items():
            assert t.dtype in sizes, (name, t.dtype)
            self.layout.append((name, t.dtype, t.shape, sizes[t.dtype]))
            sizes[t.dtype] += t.numel()
        self.buf = {
            torch.bfloat16: torch.zeros(max(1, sizes[torch.bfloat16]),
                                        dtype=torch.bfloat16,
                                        device=self.device),
            torch.float32: torch.zeros(max(1, sizes[torch.float32]),
                                       dtype=torch.float32,
                                       device=self.device),
        }
        self.ver = torch.zeros(1, dtype=torch.int32, device=self.device)

    def __getstate__(self):
        # spawn pickling ships the CUDA tensors via IPC; everything else is
        # plain metadata
        return {"layout": self.layout, "buf": self.buf, "ver": self.ver,
                "device": str(self.device)}

    def __setstate__(self, state):
        self.layout = state["layout"]
        self.buf = state["buf"]
        self.ver = state["ver"]
        self.device = torch.device(state["device"])

    # -- learner side -------------------------------------------------------
    def publish(self, pack):
        """D2D-copy the pack's tensors into the bus under a SEQLOCK: the
        version goes ODD before the copies and EVEN after, all on the
        current stream — a reader that observes an odd version (or a
        version change across its snapshot) knows a publish overlapped and
        redoes the pull.  A plain post-bump alone is NOT enough: a reader
        whose copies interleave with an in-flight publish would see a torn
        snapshot with an unchanged version (caught by
        test_weight_bus_concurrent_publish_pull_consistency)."""
        tensors = pack_tensors(pack)
        self.ver.add_(1)          # odd: publish in progress
        for name, dtype, shape, ofs in self.layout:
            t = tensors[name]
            self.buf[dtype][ofs:ofs + t.numel()].copy_(
                t.detach().view(-1), non_blocking=True)
        self.ver.add_(1)          # even: stable

    # -- actor side ---------------------------------------------------------
    def version(self) -> int:
        return int(self.ver.item())

    def pull_into(self, pack, last_ver: int) -> int:
        """If the bus moved past last_ver, copy the slices straight into
        the pack's tensors (seqlock read: retry while a publish is in
        flight or overlapped the snapshot).  Returns the (even) version
        that was applied (== last_ver if unchanged)."""
        v = self.version()
        if v == last_ver:
            return last_ver
        tensors = pack_tensors(pack)
        while True:
            if v % 2 == 1:        # publish in progress
                v = self.version()
                continue
            for name, dtype, shape, ofs in self.layout:
                t = tensors[name]
                t.view(-1).copy_(self.buf[dtype][ofs:ofs + t.numel()],
                                 non_blocking=True)
            v2 = self.version()   # .item() syncs: copies above completed
            if v2 == v:
                return v
            v = v2
