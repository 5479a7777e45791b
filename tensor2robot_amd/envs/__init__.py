"""Environment adapters (reference run_env.py:50-74 gym adapters)."""

from tensor2robot_amd.envs.adapters import GymEnvAdapter
from tensor2robot_amd.envs.adapters import TimeLimitWrapper

__all__ = ["GymEnvAdapter", "TimeLimitWrapper"]
