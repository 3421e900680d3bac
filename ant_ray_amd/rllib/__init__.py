"""ant_ray_amd.rllib — RL training on the actor runtime (focused core).

Role parity: reference rllib/ (202k LoC). This is the architectural core
at small scale, not a port: the reference's new API stack is
AlgorithmConfig -> build() -> Algorithm.train() with EnvRunner actors
collecting rollouts in parallel and a Learner updating the policy
(rllib/algorithms/ppo, rllib/env/, rllib/core/learner/). Implemented here:

  * PPOConfig / DQNConfig (environment/env_runners/training fluent API)
  * PPO.train(): N EnvRunner ACTORS collect episodes in parallel ->
    GAE advantages -> clipped-surrogate PPO update on the driver learner
    -> broadcast new weights to runners
  * CartPoleEnv: gymnasium-API built-in env (the image has no gym)

  * DQN.train(): epsilon-greedy runner actors -> uniform replay buffer ->
    double-DQN TD updates -> periodic target sync (rllib/algorithms/dqn)

Heavier reference surface (offline RL, multi-agent, DreamerV3, ...) is out
of scope for this slice.
"""
from ant_ray_amd.rllib.env import CartPoleEnv  # noqa: F401
from ant_ray_amd.rllib.dqn import DQN, DQNConfig  # noqa: F401
from ant_ray_amd.rllib.ppo import PPO, PPOConfig  # noqa: F401
