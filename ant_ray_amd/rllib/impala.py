"""IMPALA + APPO: asynchronous actor-learner algorithms.

Role parity: reference rllib/algorithms/impala/ (V-trace off-policy
correction, async sample pipelining — learner consumes batches as
runners produce them, never a synchronous barrier) and
rllib/algorithms/appo/ (IMPALA's async pipeline with PPO's clipped
surrogate on the V-trace advantages). Shares the EnvRunner/policy from
ppo.py; torch; CPU or GPU.
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

import numpy as np
import torch

from ant_ray_amd.rllib.ppo import EnvRunner, _MLPPolicy


def vtrace(behaviour_logp, target_logp, rewards, values, last_value, dones,
           gamma, lam=0.95, rho_bar=1.0, c_bar=1.0):
    """V-trace targets/advantages (IMPALA eq. 1) with the lambda discount
    on the trace coefficients (reference vtrace_torch.py `lambda_`),
    numpy, single rollout."""
    T = len(rewards)
    rho = np.minimum(np.exp(target_logp - behaviour_logp), rho_bar)
    cs = lam * np.minimum(rho, c_bar)
    next_values = np.append(values[1:], last_value)
    # a terminal step bootstraps nothing
    nonterm = 1.0 - dones.astype(np.float32)
    deltas = rho * (rewards + gamma * next_values * nonterm - values)
    vs_minus_v = np.zeros(T + 1, dtype=np.float32)
    for t in reversed(range(T)):
        vs_minus_v[t] = deltas[t] + gamma * nonterm[t] * cs[t] * vs_minus_v[t + 1]
    vs = vs_minus_v[:T] + values
    vs_next = np.append(vs[1:], last_value)
    pg_adv = rho * (rewards + gamma * vs_next * nonterm - values)
    return vs, pg_adv


class _AsyncActorLearner:
    """Common async pipeline: keep one in-flight sample() per runner;
    process whichever finishes first (reference impala.py learner thread +
    aggregation, reduced to the single-learner case)."""

    def __init__(self, cfg, loss_name):
        import ant_ray_amd as ray

        assert cfg.env_creator is not None, "call .environment() first"
        self.cfg = cfg
        if not ray.is_initialized():
            ray.init()
        probe = cfg.env_creator(0)
        self.policy = _MLPPolicy(probe.observation_dim, probe.action_dim)
        self.opt = torch.optim.Adam(self.policy.parameters(), lr=cfg.lr)
        Runner = ray.remote(EnvRunner)
        self.runners = [
            Runner.remote(cfg.env_creator, cfg.rollout_len, seed=i)
            for i in range(cfg.num_env_runners)
        ]
        self._by_ref = {}
        self.iteration = 0
        self._steps = 0
        self._loss_name = loss_name

    def _launch(self, runner):
        import ant_ray_amd as ray

        sd = {k: v.cpu() for k, v in self.policy.state_dict().items()}
        ray.get(runner.set_weights.remote(sd))
        ref = runner.sample.remote()
        self._by_ref[ref] = runner

    def train(self) -> Dict:
        """One iteration = cfg.batches_per_iter processed rollouts."""
        import ant_ray_amd as ray

        if not self._by_ref:
            for r in self.runners:
                self._launch(r)
        stats = {}
        ep_returns = []
        for _ in range(self.cfg.batches_per_iter):
            ready, _rest = ray.wait(list(self._by_ref), num_returns=1,
                                    timeout=300)
            ref = ready[0]
            runner = self._by_ref.pop(ref)
            batch = ray.get(ref)
            self._launch(runner)  # fresh weights, keep the pipeline full
            for _ in range(self.cfg.sgd_epochs):
                # v-trace is recomputed each pass: rho corrects the
                # off-policyness the earlier passes introduced
                stats = self._update(batch)
            self._steps += len(batch["obs"])
            ep_returns.extend(batch["episode_returns"])
        self.iteration += 1
        return {
            "training_iteration": self.iteration,
            "episode_return_mean": float(np.mean(ep_returns))
            if ep_returns else 0.0,
            "num_env_steps_sampled": self._steps,
            **stats,
        }

    def _update(self, batch) -> Dict:
        raise NotImplementedError

    def save(self, path: str) -> str:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save({"policy": self.policy.state_dict(),
                    "optimizer": self.opt.state_dict(),
                    "iteration": self.iteration},
                   os.path.join(path, "algorithm_state.pt"))
        return path

    def restore(self, path: str):
        import os

        st = torch.load(os.path.join(path, "algorithm_state.pt"),
                        weights_only=False)
        self.policy.load_state_dict(st["policy"])
        self.opt.load_state_dict(st["optimizer"])
        self.iteration = st["iteration"]

    def stop(self):
        import ant_ray_amd as ray

        for r in self.runners:
            try:
                ray.kill(r)
            except Exception:
                pass


class _AsyncConfig:
    def __init__(self):
        self.env_creator: Optional[Callable] = None
        self.num_env_runners = 2
        self.rollout_len = 128
        self.lr = 5e-4
        self.gamma = 0.99
        self.vf_coeff = 0.5
        self.ent_coeff = 0.01
        self.batches_per_iter = 8
        self.sgd_epochs = 4  # corrected passes per batch (rho re-weights)
        self.grad_clip = 40.0  # reference impala.py default
        self.clip = 0.3  # APPO only

    def environment(self, env_creator: Callable):
        self.env_creator = env_creator
        return self

    def env_runners(self, num_env_runners=2, rollout_fragment_length=128):
        self.num_env_runners = num_env_runners
        self.rollout_len = rollout_fragment_length
        return self

    def training(self, lr=5e-4, gamma=0.99, vf_loss_coeff=0.5,
                 entropy_coeff=0.01, batches_per_iteration=8,
                 num_sgd_iter=4, clip_param=0.3, grad_clip=40.0):
        self.lr, self.gamma = lr, gamma
        self.vf_coeff, self.ent_coeff = vf_loss_coeff, entropy_coeff
        self.batches_per_iter = batches_per_iteration
        self.sgd_epochs = num_sgd_iter
        self.clip = clip_param
        self.grad_clip = grad_clip
        return self


class IMPALAConfig(_AsyncConfig):
    def build(self) -> "IMPALA":
        return IMPALA(self)


class IMPALA(_AsyncActorLearner):
    def __init__(self, cfg: IMPALAConfig):
        super().__init__(cfg, "vtrace_pg_loss")

    def _update(self, batch) -> Dict:
        obs = torch.from_numpy(batch["obs"])
        acts = torch.from_numpy(batch["actions"])
        logits, values_now = self.policy(obs)
        dist = torch.distributions.Categorical(logits=logits)
        target_logp = dist.log_prob(acts)
        with torch.no_grad():
            # V-trace uses the LEARNER's value function at x_t and the
            # bootstrap state (the actor's value estimates are stale)
            last_v = float(self.policy(torch.from_numpy(
                batch["last_obs"]).unsqueeze(0))[1])
        vs, pg_adv = vtrace(
            batch["logp"], target_logp.detach().numpy(),
            batch["rewards"], values_now.detach().numpy(), last_v,
            batch["dones"], self.cfg.gamma)
        vs_t = torch.from_numpy(vs)
        adv_t = torch.from_numpy(pg_adv)
        adv_t = (adv_t - adv_t.mean()) / (adv_t.std() + 1e-8)
        pg_loss = -(target_logp * adv_t).mean()
        vf_loss = (values_now - vs_t).pow(2).mean()
        ent = dist.entropy().mean()
        loss = pg_loss + self.cfg.vf_coeff * vf_loss - self.cfg.ent_coeff * ent
        self.opt.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                       self.cfg.grad_clip)
        self.opt.step()
        return {"vtrace_pg_loss": float(pg_loss.detach()),
                "vf_loss": float(vf_loss.detach()),
                "entropy": float(ent.detach())}


class APPOConfig(_AsyncConfig):
    def build(self) -> "APPO":
        return APPO(self)


class APPO(_AsyncActorLearner):
    """Async PPO: IMPALA pipeline + clipped surrogate on V-trace advantages
    (reference appo_torch_policy.py)."""

    def __init__(self, cfg: APPOConfig):
        super().__init__(cfg, "appo_loss")

    def _update(self, batch) -> Dict:
        obs = torch.from_numpy(batch["obs"])
        acts = torch.from_numpy(batch["actions"])
        logp_old = torch.from_numpy(batch["logp"])
        logits, values_now = self.policy(obs)
        dist = torch.distributions.Categorical(logits=logits)
        logp = dist.log_prob(acts)
        with torch.no_grad():
            last_v = float(self.policy(torch.from_numpy(
                batch["last_obs"]).unsqueeze(0))[1])
        vs, pg_adv = vtrace(
            batch["logp"], logp.detach().numpy(),
            batch["rewards"], values_now.detach().numpy(), last_v,
            batch["dones"], self.cfg.gamma)
        adv = torch.from_numpy(pg_adv)
        adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        ratio = torch.exp(logp - logp_old)
        surr = torch.min(ratio * adv,
                         torch.clamp(ratio, 1 - self.cfg.clip,
                                     1 + self.cfg.clip) * adv)
        vf_loss = (values_now - torch.from_numpy(vs)).pow(2).mean()
        ent = dist.entropy().mean()
        loss = -surr.mean() + self.cfg.vf_coeff * vf_loss \
            - self.cfg.ent_coeff * ent
        self.opt.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                       self.cfg.grad_clip)
        self.opt.step()
        return {"appo_loss": float(loss.detach()),
                "vf_loss": float(vf_loss.detach()),
                "entropy": float(ent.detach())}
