"""PPO on CartPole (BASELINE config #1)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray_amd as ray
from ray_amd.rllib.algorithms.ppo import PPOConfig

ray.init()
algo = (
    PPOConfig()
    .environment("CartPole-v1")
    .env_runners(num_env_runners=2, num_envs_per_env_runner=8)
    .training(train_batch_size=3200, minibatch_size=256, num_epochs=8,
              entropy_coeff=0.01)
    .build()
)
for i in range(6):
    r = algo.train()
    print(f"iter {i}: reward_mean={r.get('episode_reward_mean', 0):.1f} "
          f"steps/s={r['env_steps_per_sec']:.0f}")
algo.stop()
ray.shutdown()
