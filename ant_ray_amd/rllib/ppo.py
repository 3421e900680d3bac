"""PPO on the actor runtime.

Role parity: reference rllib/algorithms/ppo/ (clipped surrogate + GAE),
rollout collection by parallel EnvRunner actors
(rllib/env/single_agent_env_runner.py), learner update + weight broadcast
(rllib/core/learner/learner_group.py). Torch policy; runs CPU or GPU.
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

import numpy as np
import torch
import torch.nn as nn


class _MLPPolicy(nn.Module):
    def __init__(self, obs_dim: int, act_dim: int, hidden: int = 64):
        super().__init__()
        self.body = nn.Sequential(
            nn.Linear(obs_dim, hidden), nn.Tanh(),
            nn.Linear(hidden, hidden), nn.Tanh(),
        )
        self.pi = nn.Linear(hidden, act_dim)
        self.vf = nn.Linear(hidden, 1)

    def forward(self, obs):
        z = self.body(obs)
        return self.pi(z), self.vf(z).squeeze(-1)


class EnvRunner:
    """Actor: owns env instances, collects rollouts with the current
    policy weights (parity single_agent_env_runner.py)."""

    def __init__(self, env_creator, rollout_len: int, seed: int):
        self.env = env_creator(seed)
        self.rollout_len = rollout_len
        self.policy = _MLPPolicy(self.env.observation_dim, self.env.action_dim)
        self.obs, _ = self.env.reset(seed=seed)
        self.episode_return = 0.0
        self.completed_returns = []

    def set_weights(self, state_dict):
        self.policy.load_state_dict(state_dict)

    def sample(self) -> Dict[str, np.ndarray]:
        obs_l, act_l, logp_l, rew_l, val_l, done_l = [], [], [], [], [], []
        with torch.no_grad():
            for _ in range(self.rollout_len):
                obs_t = torch.from_numpy(self.obs).unsqueeze(0)
                logits, value = self.policy(obs_t)
                dist = torch.distributions.Categorical(logits=logits)
                a = dist.sample()
                obs_l.append(self.obs)
                act_l.append(int(a))
                logp_l.append(float(dist.log_prob(a)))
                val_l.append(float(value))
                self.obs, r, term, trunc, _ = self.env.step(int(a))
                self.episode_return += r
                rew_l.append(r)
                done_l.append(term or trunc)
                if term or trunc:
                    self.completed_returns.append(self.episode_return)
                    self.episode_return = 0.0
                    self.obs, _ = self.env.reset()
            last_val = float(self.policy(torch.from_numpy(self.obs).unsqueeze(0))[1])
        rets = self.completed_returns[-20:]
        return {
            "obs": np.array(obs_l, dtype=np.float32),
            "actions": np.array(act_l, dtype=np.int64),
            "logp": np.array(logp_l, dtype=np.float32),
            "rewards": np.array(rew_l, dtype=np.float32),
            "values": np.array(val_l, dtype=np.float32),
            "dones": np.array(done_l, dtype=bool),
            "last_value": last_val,
            "last_obs": self.obs.copy(),  # V-trace re-values with the
            # LEARNER's value fn (impala.py); the bootstrap state travels
            "episode_returns": rets,
        }


def _gae(batch, gamma: float, lam: float):
    rew, val, done = batch["rewards"], batch["values"], batch["dones"]
    T = len(rew)
    adv = np.zeros(T, dtype=np.float32)
    last = 0.0
    next_v = batch["last_value"]
    for t in reversed(range(T)):
        nonterm = 0.0 if done[t] else 1.0
        delta = rew[t] + gamma * next_v * nonterm - val[t]
        last = delta + gamma * lam * nonterm * last
        adv[t] = last
        next_v = val[t]
    return adv, adv + val


class PPOConfig:
    """Fluent config (parity AlgorithmConfig: .environment().env_runners()
    .training().build())."""

    def __init__(self):
        self.env_creator: Optional[Callable] = None
        self.num_env_runners = 2
        self.rollout_len = 256
        self.lr = 3e-4
        self.gamma = 0.99
        self.lam = 0.95
        self.clip = 0.2
        self.epochs = 4
        self.minibatch = 128
        self.vf_coeff = 0.5
        self.ent_coeff = 0.01

    def environment(self, env_creator: Callable) -> "PPOConfig":
        self.env_creator = env_creator
        return self

    def env_runners(self, num_env_runners: int = 2,
                    rollout_fragment_length: int = 256) -> "PPOConfig":
        self.num_env_runners = num_env_runners
        self.rollout_len = rollout_fragment_length
        return self

    def training(self, lr: float = 3e-4, gamma: float = 0.99,
                 lambda_: float = 0.95, clip_param: float = 0.2,
                 num_epochs: int = 4, minibatch_size: int = 128,
                 vf_loss_coeff: float = 0.5,
                 entropy_coeff: float = 0.01) -> "PPOConfig":
        self.lr, self.gamma, self.lam, self.clip = lr, gamma, lambda_, clip_param
        self.epochs, self.minibatch = num_epochs, minibatch_size
        self.vf_coeff, self.ent_coeff = vf_loss_coeff, entropy_coeff
        return self

    def build(self) -> "PPO":
        return PPO(self)


class PPO:
    def __init__(self, config: PPOConfig):
        import ant_ray_amd as ray

        assert config.env_creator is not None, "call .environment() first"
        self.cfg = config
        if not ray.is_initialized():
            ray.init()
        probe = config.env_creator(0)
        self.policy = _MLPPolicy(probe.observation_dim, probe.action_dim)
        self.opt = torch.optim.Adam(self.policy.parameters(), lr=config.lr)
        Runner = ray.remote(EnvRunner)
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
            "policy": self.policy.state_dict(),
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
        self.policy.load_state_dict(st["policy"])
        self.opt.load_state_dict(st["optimizer"])
        self.iteration = st["iteration"]

    def train(self) -> Dict:
        """One iteration: parallel rollouts -> PPO update -> broadcast."""
        import ant_ray_amd as ray

        sd = {k: v.cpu() for k, v in self.policy.state_dict().items()}
        ray.get([r.set_weights.remote(sd) for r in self.runners])
        batches = ray.get([r.sample.remote() for r in self.runners])
        obs = torch.from_numpy(np.concatenate([b["obs"] for b in batches]))
        acts = torch.from_numpy(np.concatenate([b["actions"] for b in batches]))
        logp_old = torch.from_numpy(np.concatenate([b["logp"] for b in batches]))
        advs, rets = [], []
        for b in batches:
            a, r = _gae(b, self.cfg.gamma, self.cfg.lam)
            advs.append(a)
            rets.append(r)
        adv = torch.from_numpy(np.concatenate(advs))
        ret = torch.from_numpy(np.concatenate(rets))
        adv = (adv - adv.mean()) / (adv.std() + 1e-8)

        n = len(obs)
        idx = np.arange(n)
        stats = {}
        for _ in range(self.cfg.epochs):
            np.random.shuffle(idx)
            for s in range(0, n, self.cfg.minibatch):
                mb = idx[s:s + self.cfg.minibatch]
                logits, value = self.policy(obs[mb])
                dist = torch.distributions.Categorical(logits=logits)
                logp = dist.log_prob(acts[mb])
                ratio = torch.exp(logp - logp_old[mb])
                surr = torch.min(
                    ratio * adv[mb],
                    torch.clamp(ratio, 1 - self.cfg.clip,
                                1 + self.cfg.clip) * adv[mb])
                loss = (-surr.mean()
                        + self.cfg.vf_coeff * (value - ret[mb]).pow(2).mean()
                        - self.cfg.ent_coeff * dist.entropy().mean())
                self.opt.zero_grad()
                loss.backward()
                self.opt.step()
                stats = {"loss": float(loss)}
        self.iteration += 1
        ep_returns = [r for b in batches for r in b["episode_returns"]]
        return {
            "training_iteration": self.iteration,
            "episode_return_mean": float(np.mean(ep_returns)) if ep_returns
            else 0.0,
            "num_env_steps_sampled": n,
            **stats,
        }

    def stop(self):
        import ant_ray_amd as ray

        for r in self.runners:
            try:
                ray.kill(r)
            except Exception:
                pass
