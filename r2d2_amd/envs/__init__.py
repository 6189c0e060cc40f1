"""Environment factory (reference: /root/reference/environment.py:66).

This image has no gym/ALE/cv2, so environments are native implementations:
- 'cartpole': exact CartPole-v1 dynamics in numpy (BASELINE configs[0])
- 'synthetic': Atari-shaped synthetic frames (BASELINE benches on synthetic)
If a real `gymnasium`+ALE install is present, 'atari' wraps it with the
reference's preprocessing (grayscale, frameskip 4, 84x84 resize, noop start).
"""

from .cartpole import CartPoleEnv
from .synthetic import SyntheticAtariEnv
from .. import config as cfg


def create_env(env_type=None, seed=None, **kwargs):
    c = cfg.get()
    env_type = env_type or c.env_type
    if env_type == "cartpole":
        return CartPoleEnv(seed=seed)
    if env_type == "synthetic":
        return SyntheticAtariEnv(obs_shape=c.obs_shape, action_dim=c.action_dim,
                                 seed=seed, **kwargs)
    if env_type == "atari":
        from .atari import AtariEnv  # requires gymnasium + ALE (not in image)
        return AtariEnv(c.game_name, obs_shape=c.obs_shape, seed=seed, **kwargs)
    raise ValueError(f"unknown env_type {env_type!r}")
