"""Environment adapters for the collect/eval loop.

Reference `research/dql_grasping_lib/run_env.py:50-74`: the episode
runner accepts gym-style environments through small adapters.  The
framework's env contract (what `run_env` and `run_meta_env` call) is:

    obs = env.reset()
    obs, reward, done, debug = env.step(action)

`GymEnvAdapter` maps both gym API generations onto it (the 5-tuple
``terminated``/``truncated`` step of gymnasium and the classic
4-tuple), and `TimeLimitWrapper` bounds episode length for envs that
never set done.
"""

from __future__ import annotations

from typing import Any, Optional, Tuple

from tensor2robot_amd import gin


@gin.configurable
class GymEnvAdapter:
  """Wraps a gym/gymnasium environment into the run_env contract."""

  def __init__(self, env):
    self._env = env

  @property
  def wrapped(self):
    return self._env

  def reset(self):
    out = self._env.reset()
    if isinstance(out, tuple) and len(out) == 2:
      obs, _info = out  # gymnasium returns (obs, info)
      return obs
    return out

  def step(self, action) -> Tuple[Any, float, bool, dict]:
    out = self._env.step(action)
    if len(out) == 5:  # gymnasium: obs, reward, terminated, truncated, info
      obs, reward, terminated, truncated, info = out
      return obs, float(reward), bool(terminated or truncated), info
    obs, reward, done, info = out
    return obs, float(reward), bool(done), info

  def __getattr__(self, name):
    return getattr(self._env, name)


@gin.configurable
class TimeLimitWrapper:
  """Forces done after `max_episode_steps` steps (reference run_env's
  episode_timeout handling)."""

  def __init__(self, env, max_episode_steps: int = 100):
    self._env = env
    self._max = max_episode_steps
    self._t = 0

  def reset(self):
    self._t = 0
    return self._env.reset()

  def step(self, action):
    obs, reward, done, debug = self._env.step(action)
    self._t += 1
    if self._t >= self._max:
      done = True
    return obs, reward, done, debug

  def __getattr__(self, name):
    return getattr(self._env, name)
