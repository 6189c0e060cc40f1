"""Native CartPole-v1 dynamics (no gym dependency in this image).

Standard Barto-Sutton-Anderson cart-pole with the Gymnasium CartPole-v1
constants and termination rules: pole half-length 0.5, masspole 0.1,
masscart 1.0, force 10, dt 0.02 (Euler), terminate at |x|>2.4 or
|theta|>12 deg, truncate at 500 steps, reward 1 per step.
"""

import numpy as np


class CartPoleEnv:
    GRAVITY = 9.8
    MASSCART = 1.0
    MASSPOLE = 0.1
    LENGTH = 0.5          # half pole length
    FORCE_MAG = 10.0
    TAU = 0.02
    X_LIMIT = 2.4
    THETA_LIMIT = 12 * 2 * np.pi / 360
    MAX_STEPS = 500

    action_dim = 2
    obs_shape = (4,)

    def __init__(self, seed=None):
        self.rng = np.random.default_rng(seed)
        self.state = None
        self.steps = 0

    def reset(self) -> np.ndarray:
        self.state = self.rng.uniform(-0.05, 0.05, size=4).astype(np.float64)
        self.steps = 0
        return self.state.astype(np.float32)

    def step(self, action: int):
        x, x_dot, theta, theta_dot = self.state
        force = self.FORCE_MAG if action == 1 else -self.FORCE_MAG
        total_mass = self.MASSCART + self.MASSPOLE
        polemass_length = self.MASSPOLE * self.LENGTH
        cos_t, sin_t = np.cos(theta), np.sin(theta)
        temp = (force + polemass_length * theta_dot ** 2 * sin_t) / total_mass
        theta_acc = (self.GRAVITY * sin_t - cos_t * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * cos_t ** 2 / total_mass))
        x_acc = temp - polemass_length * theta_acc * cos_t / total_mass
        x += self.TAU * x_dot
        x_dot += self.TAU * x_acc
        theta += self.TAU * theta_dot
        theta_dot += self.TAU * theta_acc
        self.state = np.array([x, x_dot, theta, theta_dot])
        self.steps += 1
        done = bool(abs(x) > self.X_LIMIT or abs(theta) > self.THETA_LIMIT
                    or self.steps >= self.MAX_STEPS)
        return self.state.astype(np.float32), 1.0, done, {}
