"""Soft Actor-Critic (continuous actions) on the actor runtime.

Role parity: reference rllib/algorithms/sac/ (twin soft-Q + squashed
Gaussian policy + entropy temperature auto-tuning), rollout collection by
parallel EnvRunner actors, replay buffer on the learner. Torch policy;
runs CPU or GPU. Same fluent-config shape as ppo.py.
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

import numpy as np
import torch
import torch.nn as nn

LOG_STD_MIN, LOG_STD_MAX = -10.0, 2.0


class GaussianPolicy(nn.Module):
    def __init__(self, obs_dim: int, act_dim: int, hidden: int = 128):
        super().__init__()
        self.body = nn.Sequential(
            nn.Linear(obs_dim, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU(),
        )
        self.mu = nn.Linear(hidden, act_dim)
        self.log_std = nn.Linear(hidden, act_dim)

    def forward(self, obs):
        z = self.body(obs)
        return self.mu(z), self.log_std(z).clamp(LOG_STD_MIN, LOG_STD_MAX)

    def sample(self, obs):
        """Returns (tanh-squashed action, log prob)."""
        mu, log_std = self(obs)
        std = log_std.exp()
        eps = torch.randn_like(mu)
        pre = mu + std * eps
        a = torch.tanh(pre)
        logp = (-0.5 * (eps ** 2) - log_std
                - 0.5 * np.log(2 * np.pi)).sum(-1)
        # tanh change of variables
        logp = logp - torch.log(1 - a ** 2 + 1e-6).sum(-1)
        return a, logp


class QNet(nn.Module):
    def __init__(self, obs_dim: int, act_dim: int, hidden: int = 128):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(obs_dim + act_dim, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, obs, act):
        return self.net(torch.cat([obs, act], -1)).squeeze(-1)


class SACEnvRunner:
    """Actor: collects transitions with the current policy."""

    def __init__(self, env_creator, fragment: int, seed: int):
        self.env = env_creator(seed)
        self.fragment = fragment
        self.policy = GaussianPolicy(self.env.observation_dim,
                                     self.env.action_dim)
        self.obs, _ = self.env.reset(seed=seed)
        self.ep_ret = 0.0
        self.completed = []

    def set_weights(self, sd):
        self.policy.load_state_dict(sd)

    def sample(self, random_actions: bool = False) -> Dict[str, np.ndarray]:
        O, A, R, O2, D = [], [], [], [], []
        with torch.no_grad():
            for _ in range(self.fragment):
                if random_actions:
                    a = np.random.uniform(-1, 1, self.env.action_dim)
                else:
                    at, _ = self.policy.sample(
                        torch.from_numpy(self.obs).unsqueeze(0))
                    a = at.squeeze(0).numpy()
                o2, r, term, trunc, _ = self.env.step(a)
                O.append(self.obs)
                A.append(a)
                R.append(r)
                O2.append(o2)
                D.append(term)  # bootstrap through truncation
                self.ep_ret += r
                self.obs = o2
                if term or trunc:
                    self.completed.append(self.ep_ret)
                    self.ep_ret = 0.0
                    self.obs, _ = self.env.reset()
        return {
            "obs": np.array(O, dtype=np.float32),
            "actions": np.array(A, dtype=np.float32),
            "rewards": np.array(R, dtype=np.float32),
            "next_obs": np.array(O2, dtype=np.float32),
            "dones": np.array(D, dtype=bool),
            "episode_returns": self.completed[-20:],
        }


class SACConfig:
    def __init__(self):
        self.env_creator: Optional[Callable] = None
        self.num_env_runners = 2
        self.fragment = 200
        self.lr = 3e-4
        self.gamma = 0.99
        self.tau = 0.01
        self.batch = 256
        self.updates_per_iter = 64
        self.buffer_size = 100_000
        self.init_alpha = 0.2
        self.autotune_alpha = True

    def environment(self, env_creator):
        self.env_creator = env_creator
        return self

    def env_runners(self, num_env_runners=2, rollout_fragment_length=200):
        self.num_env_runners = num_env_runners
        self.fragment = rollout_fragment_length
        return self

    def training(self, lr=3e-4, gamma=0.99, tau=0.01, train_batch_size=256,
                 num_steps_sampled_before_learning_starts=0,
                 updates_per_iteration=64, initial_alpha=0.2,
                 autotune_alpha=True):
        self.lr, self.gamma, self.tau = lr, gamma, tau
        self.batch = train_batch_size
        self.updates_per_iter = updates_per_iteration
        self.init_alpha = initial_alpha
        self.autotune_alpha = autotune_alpha
        return self

    def build(self) -> "SAC":
        return SAC(self)


class SAC:
    def __init__(self, cfg: SACConfig):
        import ant_ray_amd as ray

        assert cfg.env_creator is not None
        self.cfg = cfg
        if not ray.is_initialized():
            ray.init()
        probe = cfg.env_creator(0)
        odim, adim = probe.observation_dim, probe.action_dim
        self.policy = GaussianPolicy(odim, adim)
        self.q1, self.q2 = QNet(odim, adim), QNet(odim, adim)
        self.q1_t, self.q2_t = QNet(odim, adim), QNet(odim, adim)
        self.q1_t.load_state_dict(self.q1.state_dict())
        self.q2_t.load_state_dict(self.q2.state_dict())
        self.pi_opt = torch.optim.Adam(self.policy.parameters(), lr=cfg.lr)
        self.q_opt = torch.optim.Adam(
            list(self.q1.parameters()) + list(self.q2.parameters()),
            lr=cfg.lr)
        self.log_alpha = torch.tensor(float(np.log(cfg.init_alpha)),
                                      requires_grad=True)
        self.a_opt = torch.optim.Adam([self.log_alpha], lr=cfg.lr)
        self.target_entropy = -float(adim)
        Runner = ray.remote(SACEnvRunner)
        self.runners = [Runner.remote(cfg.env_creator, cfg.fragment, seed=i)
                        for i in range(cfg.num_env_runners)]
        self._buf: Dict[str, np.ndarray] = {}
        self._n = 0
        self.iteration = 0

    def _add(self, batch):
        for k in ("obs", "actions", "rewards", "next_obs", "dones"):
            v = batch[k]
            if k not in self._buf:
                shape = (self.cfg.buffer_size,) + v.shape[1:]
                self._buf[k] = np.zeros(shape, dtype=v.dtype)
            n = len(v)
            i = self._n % self.cfg.buffer_size
            end = min(i + n, self.cfg.buffer_size)
            self._buf[k][i:end] = v[: end - i]
            if end - i < n:  # wrap
                self._buf[k][: n - (end - i)] = v[end - i:]
        self._n += len(batch["obs"])

    def _sample_buf(self, size):
        hi = min(self._n, self.cfg.buffer_size)
        idx = np.random.randint(0, hi, size=size)
        return {k: torch.from_numpy(v[idx]) for k, v in self._buf.items()}

    def train(self) -> Dict:
        import ant_ray_amd as ray

        sd = {k: v.cpu() for k, v in self.policy.state_dict().items()}
        ray.get([r.set_weights.remote(sd) for r in self.runners])
        batches = ray.get([r.sample.remote(random_actions=self._n == 0)
                           for r in self.runners])
        for b in batches:
            self._add(b)
        alpha = float(self.log_alpha.exp())
        stats = {}
        for _ in range(self.cfg.updates_per_iter):
            mb = self._sample_buf(self.cfg.batch)
            obs, act = mb["obs"], mb["actions"]
            with torch.no_grad():
                a2, logp2 = self.policy.sample(mb["next_obs"])
                qt = torch.min(self.q1_t(mb["next_obs"], a2),
                               self.q2_t(mb["next_obs"], a2))
                target = (mb["rewards"]
                          + self.cfg.gamma * (~mb["dones"]).float()
                          * (qt - alpha * logp2))
            q_loss = ((self.q1(obs, act) - target) ** 2).mean() + \
                     ((self.q2(obs, act) - target) ** 2).mean()
            self.q_opt.zero_grad()
            q_loss.backward()
            self.q_opt.step()

            a_new, logp = self.policy.sample(obs)
            q_new = torch.min(self.q1(obs, a_new), self.q2(obs, a_new))
            pi_loss = (alpha * logp - q_new).mean()
            self.pi_opt.zero_grad()
            pi_loss.backward()
            self.pi_opt.step()

            if self.cfg.autotune_alpha:
                a_loss = -(self.log_alpha
                           * (logp.detach() + self.target_entropy)).mean()
                self.a_opt.zero_grad()
                a_loss.backward()
                self.a_opt.step()
                alpha = float(self.log_alpha.exp())

            with torch.no_grad():
                for src, dst in ((self.q1, self.q1_t), (self.q2, self.q2_t)):
                    for p, pt in zip(src.parameters(), dst.parameters()):
                        pt.mul_(1 - self.cfg.tau).add_(p, alpha=self.cfg.tau)
            stats = {"q_loss": float(q_loss), "pi_loss": float(pi_loss),
                     "alpha": alpha}
        self.iteration += 1
        rets = [r for b in batches for r in b["episode_returns"]]
        return {
            "training_iteration": self.iteration,
            "episode_return_mean": float(np.mean(rets)) if rets else 0.0,
            "num_env_steps_sampled": self._n,
            **stats,
        }

    def save(self, path: str) -> str:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save({
            "policy": self.policy.state_dict(),
            "q1": self.q1.state_dict(), "q2": self.q2.state_dict(),
            "log_alpha": self.log_alpha.detach(),
            "iteration": self.iteration,
        }, os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        import os

        st = torch.load(os.path.join(path, "algorithm_state.pt"),
                        weights_only=False)
        self.policy.load_state_dict(st["policy"])
        self.q1.load_state_dict(st["q1"])
        self.q2.load_state_dict(st["q2"])
        with torch.no_grad():
            self.log_alpha.copy_(st["log_alpha"])
        self.iteration = st["iteration"]

    def stop(self):
        import ant_ray_amd as ray

        for r in self.runners:
            try:
                ray.kill(r)
            except Exception:
                pass
