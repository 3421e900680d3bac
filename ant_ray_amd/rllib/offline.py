"""Offline RL: BC and MARWIL.

Role parity: reference rllib/algorithms/bc/ (behavior cloning over an
offline dataset) and rllib/algorithms/marwil/ (monotonic advantage
re-weighted imitation learning: BC weighted by exp(beta * advantage),
with a learned value baseline). Offline data = a list of episode dicts
{"obs": [T, obs_dim], "actions": [T], "rewards": [T]} or an .npz path;
collection helper `rollout_episodes` records a policy's behavior.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Union

import numpy as np
import torch

from ant_ray_amd.rllib.ppo import _MLPPolicy

EpisodeData = List[Dict[str, np.ndarray]]


def rollout_episodes(env_creator: Callable, policy_fn: Callable,
                     n_episodes: int, seed: int = 0) -> EpisodeData:
    """Collect episodes with a scripted/learned policy (offline dataset
    generator for tests and examples)."""
    out = []
    for ep in range(n_episodes):
        env = env_creator(seed + ep)
        obs, _ = env.reset(seed=seed + ep)
        O, A, R = [], [], []
        while True:
            a = policy_fn(obs)
            O.append(obs)
            A.append(a)
            obs, r, term, trunc, _ = env.step(a)
            R.append(r)
            if term or trunc:
                break
        out.append({"obs": np.array(O, dtype=np.float32),
                    "actions": np.array(A, dtype=np.int64),
                    "rewards": np.array(R, dtype=np.float32)})
    return out


def save_episodes(path: str, episodes: EpisodeData):
    flat = {}
    for i, e in enumerate(episodes):
        for k, v in e.items():
            flat[f"ep{i}_{k}"] = v
    np.savez_compressed(path, n=len(episodes), **flat)


def load_episodes(path: str) -> EpisodeData:
    z = np.load(path)
    n = int(z["n"])
    return [{k: z[f"ep{i}_{k}"] for k in ("obs", "actions", "rewards")}
            for i in range(n)]


class _OfflineConfig:
    def __init__(self):
        self.data: Union[str, EpisodeData, None] = None
        self.obs_dim: Optional[int] = None
        self.act_dim: Optional[int] = None
        self.lr = 1e-3
        self.gamma = 0.99
        self.batch = 256
        self.updates_per_iter = 64
        self.beta = 1.0  # MARWIL advantage weight (0 -> plain BC)

    def offline_data(self, data: Union[str, EpisodeData]):
        self.data = data
        return self

    def environment(self, observation_dim: int, action_dim: int):
        self.obs_dim, self.act_dim = observation_dim, action_dim
        return self

    def training(self, lr=1e-3, gamma=0.99, train_batch_size=256,
                 updates_per_iteration=64, beta=1.0):
        self.lr, self.gamma, self.batch = lr, gamma, train_batch_size
        self.updates_per_iter = updates_per_iteration
        self.beta = beta
        return self


class BCConfig(_OfflineConfig):
    def build(self) -> "BC":
        return BC(self)


class MARWILConfig(_OfflineConfig):
    def build(self) -> "MARWIL":
        return MARWIL(self)


class BC:
    """Behavior cloning: maximize log pi(a|s) over the dataset."""

    beta = 0.0

    def __init__(self, cfg: _OfflineConfig):
        assert cfg.data is not None, "call .offline_data() first"
        episodes = (load_episodes(cfg.data) if isinstance(cfg.data, str)
                    else cfg.data)
        self.cfg = cfg
        obs_dim = cfg.obs_dim or episodes[0]["obs"].shape[1]
        act_dim = cfg.act_dim or int(
            max(e["actions"].max() for e in episodes) + 1)
        self.policy = _MLPPolicy(obs_dim, act_dim)
        self.opt = torch.optim.Adam(self.policy.parameters(), lr=cfg.lr)
        self.obs = torch.from_numpy(
            np.concatenate([e["obs"] for e in episodes]))
        self.acts = torch.from_numpy(
            np.concatenate([e["actions"] for e in episodes]))
        # monte-carlo returns for the MARWIL baseline
        rets = []
        for e in episodes:
            g, acc = np.zeros(len(e["rewards"]), dtype=np.float32), 0.0
            for t in reversed(range(len(e["rewards"]))):
                acc = e["rewards"][t] + cfg.gamma * acc
                g[t] = acc
            rets.append(g)
        ret = np.concatenate(rets)
        # standardize: the value head shares the policy body, and raw MC
        # returns (O(100) on CartPole) make the vf term dominate the shared
        # gradients and destroy the policy (measured: eval collapses to ~10)
        self._ret_mu, self._ret_sd = float(ret.mean()), float(ret.std() + 1e-8)
        self.returns = torch.from_numpy(
            (ret - self._ret_mu) / self._ret_sd)
        self.iteration = 0

    def train(self) -> Dict:
        n = len(self.obs)
        stats = {}
        for _ in range(self.cfg.updates_per_iter):
            idx = torch.randint(0, n, (self.cfg.batch,))
            logits, values = self.policy(self.obs[idx])
            dist = torch.distributions.Categorical(logits=logits)
            logp = dist.log_prob(self.acts[idx])
            beta = self.beta if self.beta is not None else self.cfg.beta
            if beta > 0:
                adv = (self.returns[idx] - values).detach()
                adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                w = torch.exp(beta * adv).clamp(max=20.0)
                pi_loss = -(w * logp).mean()
                vf_loss = (values - self.returns[idx]).pow(2).mean()
                loss = pi_loss + 0.5 * vf_loss
                stats = {"pi_loss": float(pi_loss), "vf_loss": float(vf_loss)}
            else:
                loss = -logp.mean()
                stats = {"bc_loss": float(loss)}
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()
        self.iteration += 1
        return {"training_iteration": self.iteration, **stats}

    def compute_single_action(self, obs, explore: bool = False) -> int:
        with torch.no_grad():
            logits, _ = self.policy(torch.from_numpy(
                np.asarray(obs, dtype=np.float32)).unsqueeze(0))
            if explore:
                return int(torch.distributions.Categorical(
                    logits=logits).sample())
            return int(logits.argmax())

    def evaluate(self, env_creator: Callable, episodes: int = 5,
                 seed: int = 10_000) -> float:
        total = 0.0
        for i in range(episodes):
            env = env_creator(seed + i)
            obs, _ = env.reset(seed=seed + i)
            while True:
                obs, r, term, trunc, _ = env.step(
                    self.compute_single_action(obs))
                total += r
                if term or trunc:
                    break
        return total / episodes

    def save(self, path: str) -> str:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save({"policy": self.policy.state_dict(),
                    "iteration": self.iteration},
                   os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        import os

        st = torch.load(os.path.join(path, "algorithm_state.pt"),
                        weights_only=False)
        self.policy.load_state_dict(st["policy"])
        self.iteration = st["iteration"]

    def stop(self):
        pass


class MARWIL(BC):
    """BC weighted by exp(beta * advantage) with a learned baseline."""

    beta = None  # use cfg.beta


def _flatten_transitions(episodes: EpisodeData):
    """(obs, act, rew, next_obs, done) tensors from episode dicts —
    next_obs within an episode; the final step is terminal."""
    O, A, R, NO, D = [], [], [], [], []
    for e in episodes:
        o = e["obs"]
        T = len(o)
        O.append(o)
        A.append(e["actions"])
        R.append(e["rewards"])
        no = np.concatenate([o[1:], o[-1:]], axis=0)
        NO.append(no)
        d = np.zeros(T, dtype=np.float32)
        d[-1] = 1.0
        D.append(d)
    return (torch.from_numpy(np.concatenate(O)),
            torch.from_numpy(np.concatenate(A)),
            torch.from_numpy(np.concatenate(R)),
            torch.from_numpy(np.concatenate(NO)),
            torch.from_numpy(np.concatenate(D)))


class _DiscreteQ(torch.nn.Module):
    def __init__(self, obs_dim: int, act_dim: int, hidden: int = 128):
        super().__init__()
        self.net = torch.nn.Sequential(
            torch.nn.Linear(obs_dim, hidden), torch.nn.Tanh(),
            torch.nn.Linear(hidden, hidden), torch.nn.Tanh(),
            torch.nn.Linear(hidden, act_dim))

    def forward(self, obs):
        return self.net(obs)


class CQLConfig(_OfflineConfig):
    """cql_alpha weighs the conservative regularizer
    (logsumexp_a Q(s,a) - Q(s, a_data))."""

    def __init__(self):
        super().__init__()
        self.cql_alpha = 1.0
        self.tau = 0.01

    def training(self, lr=1e-3, gamma=0.99, train_batch_size=256,
                 updates_per_iteration=64, beta=1.0, cql_alpha=1.0,
                 tau=0.01):
        super().training(lr, gamma, train_batch_size,
                         updates_per_iteration, beta)
        self.cql_alpha = cql_alpha
        self.tau = tau
        return self

    def build(self) -> "CQL":
        return CQL(self)


class CQL:
    """Conservative Q-Learning, discrete-action variant.

    Role parity: reference rllib/algorithms/cql/ (continuous SAC-CQL;
    this build's testbed envs are discrete, so the Q-learning body is
    DQN-style). TD loss on the offline transitions plus the CQL
    penalty that pushes down out-of-distribution action values:
    alpha * E[logsumexp_a Q(s,a) - Q(s, a_data)]. Acting = argmax Q.
    """

    def __init__(self, cfg: CQLConfig):
        assert cfg.data is not None, "call .offline_data() first"
        episodes = (load_episodes(cfg.data) if isinstance(cfg.data, str)
                    else cfg.data)
        self.cfg = cfg
        obs_dim = cfg.obs_dim or episodes[0]["obs"].shape[1]
        act_dim = cfg.act_dim or int(
            max(e["actions"].max() for e in episodes) + 1)
        self.q = _DiscreteQ(obs_dim, act_dim)
        self.q_t = _DiscreteQ(obs_dim, act_dim)
        self.q_t.load_state_dict(self.q.state_dict())
        self.opt = torch.optim.Adam(self.q.parameters(), lr=cfg.lr)
        (self.obs, self.acts, self.rews, self.next_obs,
         self.dones) = _flatten_transitions(episodes)
        self.iteration = 0

    def train(self) -> Dict:
        n = len(self.obs)
        stats = {}
        for _ in range(self.cfg.updates_per_iter):
            idx = torch.randint(0, n, (self.cfg.batch,))
            qs = self.q(self.obs[idx])
            q_data = qs.gather(1, self.acts[idx].unsqueeze(1)).squeeze(1)
            with torch.no_grad():
                q_next = self.q_t(self.next_obs[idx]).max(dim=1).values
                target = (self.rews[idx]
                          + self.cfg.gamma * (1 - self.dones[idx]) * q_next)
            td = (q_data - target).pow(2).mean()
            cql = (torch.logsumexp(qs, dim=1) - q_data).mean()
            loss = td + self.cfg.cql_alpha * cql
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()
            with torch.no_grad():
                for p, pt in zip(self.q.parameters(),
                                 self.q_t.parameters()):
                    pt.mul_(1 - self.cfg.tau).add_(self.cfg.tau * p)
            stats = {"td_loss": float(td), "cql_loss": float(cql)}
        self.iteration += 1
        return {"training_iteration": self.iteration, **stats}

    def compute_single_action(self, obs, explore: bool = False) -> int:
        with torch.no_grad():
            return int(self.q(torch.from_numpy(
                np.asarray(obs, dtype=np.float32)).unsqueeze(0)).argmax())

    evaluate = BC.evaluate

    def save(self, path: str) -> str:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save({"q": self.q.state_dict(),
                    "iteration": self.iteration},
                   os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        import os

        st = torch.load(os.path.join(path, "algorithm_state.pt"),
                        weights_only=False)
        self.q.load_state_dict(st["q"])
        self.q_t.load_state_dict(st["q"])
        self.iteration = st["iteration"]

    def stop(self):
        pass


class IQLConfig(_OfflineConfig):
    """expectile: the V-regression expectile (0.5 = mean, ->1 approaches
    max_a Q — the in-distribution optimal); awr_beta: advantage-weighted
    regression temperature for policy extraction."""

    def __init__(self):
        super().__init__()
        self.expectile = 0.8
        self.awr_beta = 3.0
        self.tau = 0.01

    def training(self, lr=1e-3, gamma=0.99, train_batch_size=256,
                 updates_per_iteration=64, beta=1.0, expectile=0.8,
                 awr_beta=3.0, tau=0.01):
        super().training(lr, gamma, train_batch_size,
                         updates_per_iteration, beta)
        self.expectile = expectile
        self.awr_beta = awr_beta
        self.tau = tau
        return self

    def build(self) -> "IQL":
        return IQL(self)


class IQL:
    """Implicit Q-Learning, discrete-action variant.

    Role parity: reference rllib/algorithms (IQL). Three learners, all
    strictly in-distribution (no action sampled outside the dataset):
      V  <- expectile regression towards Q_target(s, a_data)
      Q  <- TD towards r + gamma * V(s')
      pi <- advantage-weighted BC with w = exp(awr_beta * (Q - V))
    Acting = argmax pi.
    """

    def __init__(self, cfg: IQLConfig):
        assert cfg.data is not None, "call .offline_data() first"
        episodes = (load_episodes(cfg.data) if isinstance(cfg.data, str)
                    else cfg.data)
        self.cfg = cfg
        obs_dim = cfg.obs_dim or episodes[0]["obs"].shape[1]
        act_dim = cfg.act_dim or int(
            max(e["actions"].max() for e in episodes) + 1)
        self.q = _DiscreteQ(obs_dim, act_dim)
        self.q_t = _DiscreteQ(obs_dim, act_dim)
        self.q_t.load_state_dict(self.q.state_dict())
        self.v = torch.nn.Sequential(
            torch.nn.Linear(obs_dim, 128), torch.nn.Tanh(),
            torch.nn.Linear(128, 128), torch.nn.Tanh(),
            torch.nn.Linear(128, 1))
        self.policy = _MLPPolicy(obs_dim, act_dim)
        self.opt_q = torch.optim.Adam(self.q.parameters(), lr=cfg.lr)
        self.opt_v = torch.optim.Adam(self.v.parameters(), lr=cfg.lr)
        self.opt_pi = torch.optim.Adam(self.policy.parameters(), lr=cfg.lr)
        (self.obs, self.acts, self.rews, self.next_obs,
         self.dones) = _flatten_transitions(episodes)
        self.iteration = 0

    def train(self) -> Dict:
        n = len(self.obs)
        stats = {}
        for _ in range(self.cfg.updates_per_iter):
            idx = torch.randint(0, n, (self.cfg.batch,))
            obs, acts = self.obs[idx], self.acts[idx]
            with torch.no_grad():
                q_t = self.q_t(obs).gather(
                    1, acts.unsqueeze(1)).squeeze(1)
            v = self.v(obs).squeeze(1)
            diff = q_t - v
            ex = self.cfg.expectile
            v_loss = (torch.where(diff > 0, ex, 1 - ex)
                      * diff.pow(2)).mean()
            self.opt_v.zero_grad()
            v_loss.backward()
            self.opt_v.step()

            with torch.no_grad():
                v_next = self.v(self.next_obs[idx]).squeeze(1)
                target = (self.rews[idx]
                          + self.cfg.gamma * (1 - self.dones[idx]) * v_next)
            q = self.q(obs).gather(1, acts.unsqueeze(1)).squeeze(1)
            q_loss = (q - target).pow(2).mean()
            self.opt_q.zero_grad()
            q_loss.backward()
            self.opt_q.step()

            with torch.no_grad():
                adv = (self.q_t(obs).gather(1, acts.unsqueeze(1)).squeeze(1)
                       - self.v(obs).squeeze(1))
                w = torch.exp(self.cfg.awr_beta * adv).clamp(max=100.0)
            logits, _ = self.policy(obs)
            logp = torch.distributions.Categorical(
                logits=logits).log_prob(acts)
            pi_loss = -(w * logp).mean()
            self.opt_pi.zero_grad()
            pi_loss.backward()
            self.opt_pi.step()

            with torch.no_grad():
                for p, pt in zip(self.q.parameters(),
                                 self.q_t.parameters()):
                    pt.mul_(1 - self.cfg.tau).add_(self.cfg.tau * p)
            stats = {"v_loss": float(v_loss), "q_loss": float(q_loss),
                     "pi_loss": float(pi_loss)}
        self.iteration += 1
        return {"training_iteration": self.iteration, **stats}

    compute_single_action = BC.compute_single_action
    evaluate = BC.evaluate

    def save(self, path: str) -> str:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save({"policy": self.policy.state_dict(),
                    "q": self.q.state_dict(),
                    "iteration": self.iteration},
                   os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        import os

        st = torch.load(os.path.join(path, "algorithm_state.pt"),
                        weights_only=False)
        self.policy.load_state_dict(st["policy"])
        self.q.load_state_dict(st["q"])
        self.iteration = st["iteration"]

    def stop(self):
        pass
