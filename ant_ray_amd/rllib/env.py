"""Built-in envs with the gymnasium step/reset API (no gym in this image).

Parity note: the reference wraps gymnasium envs in EnvRunner
(rllib/env/single_agent_env_runner.py); any object with the same
reset()/step() contract works here.
"""
from __future__ import annotations

import numpy as np


class CartPoleEnv:
    """Classic CartPole-v1 dynamics (gymnasium-compatible API)."""

    observation_dim = 4
    action_dim = 2

    def __init__(self, seed=None, max_steps: int = 500):
        self.rng = np.random.RandomState(seed)
        self.max_steps = max_steps
        self.state = None
        self.t = 0

    def reset(self, *, seed=None, options=None):
        if seed is not None:
            self.rng = np.random.RandomState(seed)
        self.state = self.rng.uniform(-0.05, 0.05, size=4).astype(np.float32)
        self.t = 0
        return self.state.copy(), {}

    def step(self, action: int):
        x, x_dot, th, th_dot = self.state
        force = 10.0 if action == 1 else -10.0
        costh, sinth = np.cos(th), np.sin(th)
        temp = (force + 0.05 * th_dot**2 * sinth) / 1.1
        th_acc = (9.8 * sinth - costh * temp) / (0.5 * (4.0 / 3.0 - 0.1 * costh**2 / 1.1))
        x_acc = temp - 0.05 * th_acc * costh / 1.1
        tau = 0.02
        self.state = np.array([
            x + tau * x_dot, x_dot + tau * x_acc,
            th + tau * th_dot, th_dot + tau * th_acc,
        ], dtype=np.float32)
        self.t += 1
        terminated = bool(abs(self.state[0]) > 2.4 or abs(self.state[2]) > 0.2095)
        truncated = self.t >= self.max_steps
        return self.state.copy(), 1.0, terminated, truncated, {}


class Reacher1DEnv:
    """Continuous-control smoke env: drive a 1-D point mass to a target.
    obs = [pos, vel, target]; action in [-1, 1]; reward = -|pos - target|
    - 0.01*a^2. Learnable by SAC in a few thousand steps."""

    observation_dim = 3
    action_dim = 1
    continuous = True

    def __init__(self, seed=None, max_steps: int = 60):
        self.rng = np.random.RandomState(seed)
        self.max_steps = max_steps
        self.pos = self.vel = self.target = 0.0
        self.t = 0

    def _obs(self):
        return np.array([self.pos, self.vel, self.target], dtype=np.float32)

    def reset(self, *, seed=None, options=None):
        if seed is not None:
            self.rng = np.random.RandomState(seed)
        self.pos = float(self.rng.uniform(-1, 1))
        self.vel = 0.0
        self.target = float(self.rng.uniform(-1, 1))
        self.t = 0
        return self._obs(), {}

    def step(self, action):
        a = float(np.clip(np.asarray(action).reshape(-1)[0], -1, 1))
        self.vel = 0.8 * self.vel + 0.2 * a
        self.pos = float(np.clip(self.pos + 0.2 * self.vel, -2, 2))
        self.t += 1
        reward = -abs(self.pos - self.target) - 0.01 * a * a
        truncated = self.t >= self.max_steps
        return self._obs(), reward, False, truncated, {}
