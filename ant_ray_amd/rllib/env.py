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
