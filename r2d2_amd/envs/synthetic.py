"""Synthetic Atari-shaped environment.

Stands in for ALE (no network / no ALE ROMs in this image; the reference
wraps `gym.make('ALE/MsPacman-v5')`, /root/reference/environment.py:66-74).
Produces uint8 frames of the configured (C, 84, 84) shape with a cheap
deterministic-per-seed structure: a random per-episode base pattern plus a
moving blob whose position depends on the action history, episodic reward
with sparse positive events, and geometric episode termination.  The shape,
dtype, and timing behavior match what the data path must sustain; the
"game" itself is synthetic (BASELINE.json benches on synthetic frames).
"""

import numpy as np


class SyntheticAtariEnv:
    def __init__(self, obs_shape=(4, 84, 84), action_dim=9, mean_ep_len=600,
                 seed=None):
        self.obs_shape = tuple(obs_shape)
        self.action_dim = action_dim
        self.mean_ep_len = mean_ep_len
        self.rng = np.random.default_rng(seed)
        self._base = None
        self._pos = None
        self.steps = 0

    def _frame(self) -> np.ndarray:
        f = self._base.copy()
        c, h, w = self.obs_shape
        y, x = int(self._pos[0]) % (h - 8), int(self._pos[1]) % (w - 8)
        f[:, y:y + 8, x:x + 8] = 255
        return f

    def reset(self) -> np.ndarray:
        c, h, w = self.obs_shape
        self._base = (self.rng.integers(0, 64, size=(c, h, w))).astype(np.uint8)
        self._pos = np.array([h // 2, w // 2], dtype=np.int64)
        self.steps = 0
        return self._frame()

    def step(self, action: int):
        moves = [(-2, 0), (2, 0), (0, -2), (0, 2)]
        dy, dx = moves[action % 4]
        self._pos += (dy, dx)
        self.steps += 1
        reward = float(self.rng.random() < 0.05) * 10.0
        done = bool(self.rng.random() < 1.0 / self.mean_ep_len)
        return self._frame(), reward, done, {}
