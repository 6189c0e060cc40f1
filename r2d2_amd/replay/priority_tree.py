"""CPU prioritized-replay sum-tree (numpy), level-array layout.

Semantics match the reference PriorityTree (/root/reference/priority_tree.py):
- priorities = td^alpha stored at leaves, capacity rounded up to a power of 2
- update: vectorized leaf write + level-by-level parent re-sum
- sample: stratified — the total mass is split into n equal intervals with a
  uniform jitter inside each; vectorized n-way descent
- IS weights = (p / min_p)^-beta, normalized by the BATCH min (no 1/N factor)

The implementation differs structurally (per-level arrays instead of one flat
heap; the GPU twin in ops/hip/sumtree.hip uses the same per-level layout so
both sides share indexing math).
"""

from typing import Optional, Tuple

import numpy as np


class PriorityTree:
    def __init__(self, capacity: int, prio_exponent: float, is_exponent: float,
                 rng: Optional[np.random.Generator] = None):
        self.capacity = capacity
        self.num_leaves = 1
        while self.num_leaves < capacity:
            self.num_leaves *= 2
        # levels[0] is the root; levels[-1] are the leaves
        self.levels = []
        n = 1
        while n <= self.num_leaves:
            self.levels.append(np.zeros(n, dtype=np.float64))
            n *= 2
        self.prio_exponent = prio_exponent
        self.is_exponent = is_exponent
        self.rng = rng or np.random.default_rng()

    @property
    def total(self) -> float:
        return float(self.levels[0][0])

    def update(self, idxes: np.ndarray, td_error: np.ndarray) -> None:
        idxes = np.asarray(idxes, dtype=np.int64)
        prios = np.asarray(td_error, dtype=np.float64) ** self.prio_exponent
        self.levels[-1][idxes] = prios
        child = self.levels[-1]
        nodes = idxes
        for lvl in range(len(self.levels) - 2, -1, -1):
            nodes = np.unique(nodes >> 1)
            self.levels[lvl][nodes] = child[2 * nodes] + child[2 * nodes + 1]
            child = self.levels[lvl]

    def leaf_values(self) -> np.ndarray:
        """Copy of the raw stored leaf priorities (already ^alpha) — the
        snapshot format for replay persistence."""
        return self.levels[-1].copy()

    def set_leaf_values(self, leaves: np.ndarray) -> None:
        """Restore raw leaf priorities (as returned by leaf_values — NOT
        td errors; no ^alpha is applied) and rebuild every parent level."""
        if leaves.shape != self.levels[-1].shape:
            raise ValueError(f"leaf snapshot shape {leaves.shape} != tree "
                             f"leaves {self.levels[-1].shape}")
        self.levels[-1][:] = leaves
        child = self.levels[-1]
        for lvl in range(len(self.levels) - 2, -1, -1):
            self.levels[lvl][:] = child[0::2] + child[1::2]
            child = self.levels[lvl]

    def sample(self, num_samples: int) -> Tuple[np.ndarray, np.ndarray]:
        total = self.levels[0][0]
        assert total > 0, "sampling from an empty tree"
        # stratified prefix targets
        u = (np.arange(num_samples, dtype=np.float64)
             + self.rng.uniform(0.0, 1.0, num_samples)) * (total / num_samples)
        idxes = np.zeros(num_samples, dtype=np.int64)
        for lvl in range(1, len(self.levels)):
            left = self.levels[lvl][2 * idxes]
            go_right = u >= left
            u = np.where(go_right, u - left, u)
            idxes = 2 * idxes + go_right
        priorities = self.levels[-1][idxes]
        # fp-edge repair: accumulated rounding in the parent sums can steer
        # a descent into a ZERO-priority leaf (zero-padded capacity tail,
        # or the unused slots of a partial block).  Snap such samples to
        # the nearest preceding nonzero leaf — the stratum's true mass
        # lives at/left of the overshoot.  (The consumer indexes blocks
        # with these, so returning a dead slot is a crash, not just a
        # skewed IS weight.)
        bad = priorities <= 0.0
        if bad.any():
            valid = np.flatnonzero(self.levels[-1])
            pos = np.searchsorted(valid, idxes[bad], side="right") - 1
            idxes[bad] = valid[np.clip(pos, 0, len(valid) - 1)]
            priorities = self.levels[-1][idxes]
        min_p = priorities.min() if len(priorities) else 1.0
        is_weights = np.power(np.maximum(priorities, min_p * 1e-12) / min_p,
                              -self.is_exponent)
        return idxes, is_weights
