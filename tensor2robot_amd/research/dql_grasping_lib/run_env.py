"""Episode runner: policy <-> env interaction + replay writing.

Reference: `research/dql_grasping_lib/run_env.py:77-235` — episode loop:
reset, explore schedule, policy.sample_action, env.step, accumulate
(s, a, r, s', done) transitions, episode_to_transitions_fn ->
replay_writer.write, episode-reward summaries.
"""

from __future__ import annotations

import collections
import logging
import os
from typing import Callable, Optional


from tensor2robot_amd import gin
from tensor2robot_amd.utils import summaries as summaries_mod

_log = logging.getLogger(__name__)

Transition = collections.namedtuple(
    "Transition", ["state", "action", "reward", "next_state", "done",
                   "debug"])


@gin.configurable
def linear_explore_schedule(step: int, initial: float = 1.0,
                            final: float = 0.1,
                            decay_steps: int = 100000) -> float:
  frac = min(max(step, 0) / float(decay_steps), 1.0)
  return initial + frac * (final - initial)


@gin.configurable
def episode_to_transitions_identity(episode_data):
  """Default episode -> serialized records fn: caller supplies bytes."""
  return [t for t in episode_data]


@gin.configurable
def run_env(env, policy=None, explore_schedule=None, episode_to_transitions_fn=None,
            replay_writer=None, root_dir: Optional[str] = None,
            task: int = 0, global_step: int = 0, num_episodes: int = 1,
            tag: str = "collect", max_episode_steps: Optional[int] = None):
  """Runs episodes; returns list of per-episode total rewards.

  Reference :77-235.  `episode_to_transitions_fn` maps the episode's
  Transition list to serialized records for the replay writer.
  """
  writer_path = None
  summary_writer = None
  if root_dir:
    os.makedirs(root_dir, exist_ok=True)
    summary_writer = summaries_mod.SummaryWriter(
        os.path.join(root_dir, f"live_eval_{task}"))
  if replay_writer is not None and root_dir:
    writer_path = os.path.join(root_dir, "policy_collect",
                               f"task{task}_step{global_step}")
    os.makedirs(os.path.dirname(writer_path), exist_ok=True)
    replay_writer.open(writer_path)

  episode_rewards = []
  try:
    for episode in range(num_episodes):
      policy.reset()
      obs = env.reset()
      episode_data = []
      total_reward = 0.0
      done = False
      step = 0
      explore_prob = explore_schedule(global_step) if explore_schedule \
          else 0.0
      while not done:
        action, debug = policy.sample_action(obs, explore_prob)
        next_obs, reward, done, env_debug = env.step(action)
        episode_data.append(Transition(obs, action, reward, next_obs,
                                       done, env_debug))
        total_reward += float(reward)
        obs = next_obs
        step += 1
        if max_episode_steps is not None and step >= max_episode_steps:
          break
      episode_rewards.append(total_reward)
      if summary_writer is not None:
        summary_writer.add_scalar(f"{tag}/episode_reward", total_reward,
                                  global_step + episode)
      if replay_writer is not None and episode_to_transitions_fn:
        records = episode_to_transitions_fn(episode_data)
        for rec in records:
          replay_writer.write(rec)
  finally:
    if replay_writer is not None and writer_path is not None:
      replay_writer.close()
    if summary_writer is not None:
      summary_writer.close()
  return episode_rewards


def encode_image_array_as_png_str(image) -> bytes:
  """uint8 HWC array -> PNG bytes (reference run_env.py helper)."""
  import numpy as _np
  from tensor2robot_amd.data import image_codec
  return image_codec.encode_png(_np.asarray(image, _np.uint8))
