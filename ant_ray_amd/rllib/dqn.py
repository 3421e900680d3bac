"""DQN on the actor runtime.

Role parity: reference rllib/algorithms/dqn/ (double-DQN target, replay
buffer rllib/utils/replay_buffers/, epsilon-greedy exploration) with the
same EnvRunner-actor rollout architecture as ppo.py: parallel sampling
actors -> driver learner -> weight broadcast.
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

import numpy as np
import torch
import torch.nn as nn


class _QNet(nn.Module):
    def __init__(self, obs_dim: int, act_dim: int, hidden: int = 64):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(obs_dim, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU(),
            nn.Linear(hidden, act_dim),
        )

    def forward(self, obs):
        return self.net(obs)


class QEnvRunner:
    """Actor: epsilon-greedy rollouts with the current Q-network."""

    def __init__(self, env_creator, rollout_len: int, seed: int):
        self.env = env_creator(seed)
        self.rollout_len = rollout_len
        self.q = _QNet(self.env.observation_dim, self.env.action_dim)
        self.rng = np.random.RandomState(seed)
        self.obs, _ = self.env.reset(seed=seed)
        self.episode_return = 0.0
        self.completed_returns = []

    def set_weights(self, state_dict):
        self.q.load_state_dict(state_dict)

    def sample(self, epsilon: float) -> Dict[str, np.ndarray]:
        obs_l, act_l, rew_l, next_l, done_l = [], [], [], [], []
        with torch.no_grad():
            for _ in range(self.rollout_len):
                if self.rng.rand() < epsilon:
                    a = int(self.rng.randint(self.env.action_dim))
                else:
                    qv = self.q(torch.from_numpy(self.obs).unsqueeze(0))
                    a = int(qv.argmax(-1))
                obs_l.append(self.obs)
                act_l.append(a)
                self.obs, r, term, trunc, _ = self.env.step(a)
                self.episode_return += r
                rew_l.append(r)
                next_l.append(self.obs)
                done_l.append(term)  # bootstrap through truncation
                if term or trunc:
                    self.completed_returns.append(self.episode_return)
                    self.episode_return = 0.0
                    self.obs, _ = self.env.reset()
        return {
            "obs": np.array(obs_l, dtype=np.float32),
            "actions": np.array(act_l, dtype=np.int64),
            "rewards": np.array(rew_l, dtype=np.float32),
            "next_obs": np.array(next_l, dtype=np.float32),
            "dones": np.array(done_l, dtype=bool),
            "episode_returns": self.completed_returns[-20:],
        }


class _ReplayBuffer:
    """Uniform FIFO replay (parity: utils/replay_buffers/
    episode_replay_buffer.py at transition granularity)."""

    def __init__(self, capacity: int, obs_dim: int):
        self.capacity = capacity
        self.obs = np.zeros((capacity, obs_dim), dtype=np.float32)
        self.next_obs = np.zeros((capacity, obs_dim), dtype=np.float32)
        self.actions = np.zeros(capacity, dtype=np.int64)
        self.rewards = np.zeros(capacity, dtype=np.float32)
        self.dones = np.zeros(capacity, dtype=bool)
        self.size = 0
        self.pos = 0

    def add_batch(self, b: Dict[str, np.ndarray]):
        n = len(b["obs"])
        for i in range(n):
            p = self.pos
            self.obs[p] = b["obs"][i]
            self.next_obs[p] = b["next_obs"][i]
            self.actions[p] = b["actions"][i]
            self.rewards[p] = b["rewards"][i]
            self.dones[p] = b["dones"][i]
            self.pos = (p + 1) % self.capacity
            self.size = min(self.size + 1, self.capacity)

    def sample(self, n: int, rng: np.random.RandomState):
        idx = rng.randint(0, self.size, size=n)
        return (self.obs[idx], self.actions[idx], self.rewards[idx],
                self.next_obs[idx], self.dones[idx])


class DQNConfig:
    """Fluent config (parity AlgorithmConfig)."""

    def __init__(self):
        self.env_creator: Optional[Callable] = None
        self.num_env_runners = 2
        self.rollout_len = 128
        self.lr = 1e-3
        self.gamma = 0.99
        self.buffer_size = 50_000
        self.batch_size = 64
        self.updates_per_iter = 32
        self.target_update_freq = 2  # iterations between target syncs
        self.eps_start = 1.0
        self.eps_end = 0.05
        self.eps_decay_iters = 20
        self.double_q = True
        self.seed = 0

    def environment(self, env_creator: Callable) -> "DQNConfig":
        self.env_creator = env_creator
        return self

    def env_runners(self, num_env_runners: int = 2,
                    rollout_fragment_length: int = 128) -> "DQNConfig":
        self.num_env_runners = num_env_runners
        self.rollout_len = rollout_fragment_length
        return self

    def training(self, lr: float = 1e-3, gamma: float = 0.99,
                 buffer_size: int = 50_000, batch_size: int = 64,
                 updates_per_iter: int = 32, target_update_freq: int = 2,
                 double_q: bool = True) -> "DQNConfig":
        self.lr, self.gamma = lr, gamma
        self.buffer_size, self.batch_size = buffer_size, batch_size
        self.updates_per_iter = updates_per_iter
        self.target_update_freq = target_update_freq
        self.double_q = double_q
        return self

    def exploration(self, eps_start: float = 1.0, eps_end: float = 0.05,
                    eps_decay_iters: int = 20) -> "DQNConfig":
        self.eps_start, self.eps_end = eps_start, eps_end
        self.eps_decay_iters = eps_decay_iters
        return self

    def build(self) -> "DQN":
        return DQN(self)


class DQN:
    def __init__(self, config: DQNConfig):
        import ant_ray_amd as ray

        assert config.env_creator is not None, "call .environment() first"
        self.cfg = config
        if not ray.is_initialized():
            ray.init()
        probe = config.env_creator(0)
        self.q = _QNet(probe.observation_dim, probe.action_dim)
        self.target = _QNet(probe.observation_dim, probe.action_dim)
        self.target.load_state_dict(self.q.state_dict())
        self.opt = torch.optim.Adam(self.q.parameters(), lr=config.lr)
        self.buffer = _ReplayBuffer(config.buffer_size, probe.observation_dim)
        self.rng = np.random.RandomState(config.seed)
        Runner = ray.remote(QEnvRunner)
        self.runners = [
            Runner.remote(config.env_creator, config.rollout_len, seed=i)
            for i in range(config.num_env_runners)
        ]
        self.iteration = 0


    def save(self, path: str) -> str:
        """Persist the algorithm state (parity: Algorithm.save_checkpoint)."""
        import os

        import torch as _torch

        os.makedirs(path, exist_ok=True)
        _torch.save({
            "policy": self.q.state_dict(),
            "optimizer": self.opt.state_dict(),
            "iteration": self.iteration,
        }, os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        """Load state saved by save() (parity: Algorithm.restore)."""
        import os

        import torch as _torch

        st = _torch.load(os.path.join(path, "algorithm_state.pt"),
                         weights_only=False)
        self.q.load_state_dict(st["policy"])
        self.target.load_state_dict(st["policy"])
        self.opt.load_state_dict(st["optimizer"])
        self.iteration = st["iteration"]

    def _epsilon(self) -> float:
        c = self.cfg
        frac = min(1.0, self.iteration / max(1, c.eps_decay_iters))
        return c.eps_start + (c.eps_end - c.eps_start) * frac

    def train(self) -> Dict:
        """One iteration: parallel epsilon-greedy rollouts -> replay ->
        double-DQN TD updates -> periodic target sync -> broadcast."""
        import ant_ray_amd as ray

        sd = {k: v.cpu() for k, v in self.q.state_dict().items()}
        ray.get([r.set_weights.remote(sd) for r in self.runners])
        eps = self._epsilon()
        batches = ray.get([r.sample.remote(eps) for r in self.runners])
        for b in batches:
            self.buffer.add_batch(b)

        stats = {"loss": None}
        if self.buffer.size >= self.cfg.batch_size:
            losses = []
            for _ in range(self.cfg.updates_per_iter):
                obs, act, rew, nxt, done = self.buffer.sample(
                    self.cfg.batch_size, self.rng)
                obs_t = torch.from_numpy(obs)
                nxt_t = torch.from_numpy(nxt)
                with torch.no_grad():
                    if self.cfg.double_q:
                        best = self.q(nxt_t).argmax(-1, keepdim=True)
                        next_q = self.target(nxt_t).gather(1, best).squeeze(1)
                    else:
                        next_q = self.target(nxt_t).max(-1).values
                    tgt = (torch.from_numpy(rew)
                           + self.cfg.gamma * next_q
                           * (~torch.from_numpy(done)).float())
                qv = self.q(obs_t).gather(
                    1, torch.from_numpy(act).unsqueeze(1)).squeeze(1)
                loss = nn.functional.smooth_l1_loss(qv, tgt)
                self.opt.zero_grad()
                loss.backward()
                self.opt.step()
                losses.append(float(loss))
            stats["loss"] = float(np.mean(losses))

        self.iteration += 1
        if self.iteration % self.cfg.target_update_freq == 0:
            self.target.load_state_dict(self.q.state_dict())
        ep_returns = [r for b in batches for r in b["episode_returns"]]
        return {
            "training_iteration": self.iteration,
            "epsilon": eps,
            "episode_return_mean": (float(np.mean(ep_returns))
                                    if ep_returns else float("nan")),
            "replay_size": self.buffer.size,
            **stats,
        }

    def stop(self):
        import ant_ray_amd as ray

        for r in self.runners:
            try:
                ray.kill(r)
            except Exception:
                pass
