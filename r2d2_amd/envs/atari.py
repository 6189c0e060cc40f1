"""Optional ALE wrapper with the reference's preprocessing semantics
(/root/reference/environment.py): grayscale, frameskip 4, no repeat-action
stochasticity, 1-30 random noops at reset, area-resize to 84x84 uint8.

Requires `gymnasium` + ALE ROMs, which this image does not ship; import is
deferred so the rest of the package works without them.  The resize is a
numpy box-filter (no cv2 in the image).
"""

from functools import lru_cache

import numpy as np

_blas_limiter = None


def limit_blas_threads():
    """Pin BLAS pools to ONE thread in this process (idempotent, kept for
    the process lifetime).  Actor processes are single-threaded by design
    (the reference sets torch.set_num_threads(1), train.py:13) but numpy's
    OpenBLAS pool is separate: on a saturated machine the multithreaded
    pool collapses on the tiny per-frame resize matmuls (measured 28
    frames/s loaded vs 6,078 single-threaded on 8 contended cores —
    thread-pool handoff costs 200x the 3 MFLOP of work)."""
    global _blas_limiter
    if _blas_limiter is None:
        try:
            from threadpoolctl import threadpool_limits
            _blas_limiter = threadpool_limits(limits=1)
        except Exception:      # threadpoolctl absent: leave pools alone
            _blas_limiter = False


@lru_cache(maxsize=8)
def _area_weights(n_in: int, n_out: int) -> np.ndarray:
    """(n_out, n_in) row-stochastic matrix of fractional pixel-area overlaps:
    output cell i covers input span [i*s, (i+1)*s), s = n_in/n_out, with
    partial pixels weighted by their covered fraction — the "pixel area
    relation" cv2.INTER_AREA computes (reference environment.py:55-63 resizes
    210x160 -> 84x84, a FRACTIONAL 2.5x/1.905x ratio)."""
    s = n_in / n_out
    edges = np.arange(n_out + 1) * s
    r = np.arange(n_in)
    lo = np.maximum(edges[:-1, None], r[None, :])
    hi = np.minimum(edges[1:, None], r[None, :] + 1.0)
    w = np.clip(hi - lo, 0.0, None)
    return (w / s).astype(np.float64)


def _area_resize_84(img: np.ndarray) -> np.ndarray:
    """Exact area-average resize HxW -> 84x84 as two small matrix products
    (no per-pixel Python loop; ~8 MFLOP/frame -> >10k frames/s per core)."""
    h, w = img.shape
    wy = _area_weights(h, 84)          # (84, H)
    wx = _area_weights(w, 84)          # (84, W)
    out = wy @ img.astype(np.float64) @ wx.T
    return np.clip(np.rint(out), 0, 255).astype(np.uint8)


class AtariEnv:
    def __init__(self, game_name: str, obs_shape=(1, 84, 84), noop_max: int = 30,
                 seed=None):
        import gymnasium as gym
        limit_blas_threads()   # this process does per-frame resize matmuls
        self.env = gym.make(f"ALE/{game_name}-v5", obs_type="grayscale",
                            frameskip=4, repeat_action_probability=0)
        self.action_dim = self.env.action_space.n
        self.obs_shape = obs_shape
        self.noop_max = noop_max
        self.rng = np.random.default_rng(seed)
        self._stack = None

    def _warp(self, frame: np.ndarray) -> np.ndarray:
        small = _area_resize_84(frame)
        c = self.obs_shape[0]
        if c == 1:
            return small[None]
        if self._stack is None:
            self._stack = np.repeat(small[None], c, axis=0)
        else:
            self._stack = np.concatenate([self._stack[1:], small[None]], axis=0)
        return self._stack.copy()

    def reset(self) -> np.ndarray:
        self._stack = None
        obs, _ = self.env.reset()
        for _ in range(int(self.rng.integers(1, self.noop_max + 1))):
            obs, _, term, trunc, _ = self.env.step(0)
            if term or trunc:
                obs, _ = self.env.reset()
        return self._warp(obs)

    def step(self, action: int):
        obs, reward, term, trunc, info = self.env.step(action)
        return self._warp(obs), float(reward), bool(term or trunc), info
