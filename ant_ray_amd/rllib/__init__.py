"""ant_ray_amd.rllib — RL training on the actor runtime (focused core).

Role parity: reference rllib/ (202k LoC). This is the architectural core
at small scale, not a port: the reference's new API stack is
AlgorithmConfig -> build() -> Algorithm.train() with EnvRunner actors
collecting rollouts in parallel and a Learner updating the policy
(rllib/algorithms/ppo, rllib/env/, rllib/core/learner/). Implemented here
(each a fluent Config -> build() -> .train()/.save()/.restore()):

  * PPO   — parallel EnvRunner actors -> GAE -> clipped surrogate
  * DQN   — epsilon-greedy runners -> replay -> double-DQN + target sync
  * SAC   — continuous control: squashed-Gaussian policy, twin soft-Q,
            entropy temperature auto-tuning, replay (algorithms/sac)
  * IMPALA— async actor-learner with V-trace off-policy correction
            (ray.wait pipeline, never a synchronous barrier)
  * APPO  — IMPALA's async pipeline + PPO's clipped surrogate
  * BC    — offline behavior cloning (algorithms/bc)
  * MARWIL— advantage-weighted imitation with learned baseline
  * Built-in envs: CartPoleEnv (discrete), Reacher1DEnv (continuous) —
    gymnasium step/reset API (the image has no gym)

Heavier reference surface (multi-agent, DreamerV3, connectors) remains
out of scope.
"""
from ant_ray_amd.rllib.env import CartPoleEnv, Reacher1DEnv  # noqa: F401
from ant_ray_amd.rllib.dqn import DQN, DQNConfig  # noqa: F401
from ant_ray_amd.rllib.ppo import PPO, PPOConfig  # noqa: F401
from ant_ray_amd.rllib.sac import SAC, SACConfig  # noqa: F401
from ant_ray_amd.rllib.impala import (  # noqa: F401
    APPO,
    APPOConfig,
    IMPALA,
    IMPALAConfig,
)
from ant_ray_amd.rllib.offline import (  # noqa: F401
    BC,
    BCConfig,
    CQL,
    CQLConfig,
    IQL,
    IQLConfig,
    MARWIL,
    MARWILConfig,
    load_episodes,
    rollout_episodes,
    save_episodes,
)
