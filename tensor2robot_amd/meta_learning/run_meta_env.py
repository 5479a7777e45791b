"""Meta-learning env interaction loop: demo -> adapt -> trial cycles.

Reference `meta_learning/run_meta_env.py:32-...`: per task — reset task,
collect num_demos demonstration episodes (or pull env.task_data),
policy.adapt(condition_data), then num_adaptations_per_task rounds of
num_episodes_per_adaptation trial episodes, re-adapting between rounds
with the accumulated episodes; transitions written per task via the
replay writer; per-adaptation-step average rewards summarized.
"""

from __future__ import annotations

import collections
import copy
import logging
import os
from typing import Optional

import numpy as np

from tensor2robot_amd import gin
from tensor2robot_amd.utils import summaries as summaries_mod

_log = logging.getLogger(__name__)


@gin.configurable
def run_meta_env(env, policy=None, demo_policy_cls=None,
                 explore_schedule=None, episode_to_transitions_fn=None,
                 replay_writer=None, root_dir: Optional[str] = None,
                 task: int = 0, global_step: int = 0,
                 num_episodes=None,  # accepted, unused (reference :41)
                 num_tasks: int = 10, num_adaptations_per_task: int = 2,
                 num_episodes_per_adaptation: int = 1, num_demos: int = 1,
                 break_after_one_task: bool = False, tag: str = "collect",
                 max_episode_steps: Optional[int] = None):
  """Runs the meta agent+env loop; returns per-step mean rewards."""
  task_step_rewards = collections.defaultdict(
      lambda: collections.defaultdict(list))
  summary_writer = None
  if root_dir:
    os.makedirs(root_dir, exist_ok=True)
    summary_writer = summaries_mod.SummaryWriter(
        os.path.join(root_dir, f"live_eval_{task}"))

  def _run_demo():
    obs = env.reset()
    demo_policy = demo_policy_cls(env)
    episode_data = []
    while True:
      action, debug = demo_policy.sample_action(obs, 0)
      if action is None:
        break
      next_obs, rew, done, env_debug = env.step(action)
      env_debug = dict(env_debug or {})
      env_debug["is_demo"] = True
      episode_data.append((obs, action, rew, next_obs, done, env_debug))
      obs = next_obs
      if done:
        break
    return episode_data

  for task_idx in range(num_tasks):
    if hasattr(policy, "reset_task"):
      policy.reset_task()
    if hasattr(env, "reset_task"):
      env.reset_task()
    if replay_writer is not None and root_dir:
      record_name = os.path.join(
          root_dir, f"gs{global_step}_t{task}_{task_idx}")
      replay_writer.open(record_name)

    condition_data = []
    if demo_policy_cls is not None and hasattr(policy, "adapt"):
      for _ in range(num_demos):
        episode_data = _run_demo()
        condition_data.append(episode_data)
        if replay_writer is not None and episode_to_transitions_fn:
          for rec in episode_to_transitions_fn(episode_data):
            replay_writer.write(rec)
      policy.adapt(copy.copy(condition_data))
    elif hasattr(env, "task_data") and hasattr(policy, "adapt"):
      for episode_name, episode_data in env.task_data.items():
        if str(episode_name).startswith("condition_ep"):
          condition_data.append(episode_data)
      policy.adapt(copy.copy(condition_data))

    for step_num in range(num_adaptations_per_task):
      if step_num != 0 and hasattr(policy, "adapt"):
        policy.adapt(copy.copy(condition_data))
      for _ in range(num_episodes_per_adaptation):
        done, env_step, episode_reward, episode_data = False, 0, 0.0, []
        policy.reset()
        obs = env.reset()
        explore_prob = explore_schedule(global_step) if explore_schedule \
            else 0.0
        while not done:
          action, policy_debug = policy.sample_action(obs, explore_prob)
          new_obs, rew, done, env_debug = env.step(action)
          debug = dict(policy_debug or {})
          debug.update(env_debug or {})
          env_step += 1
          episode_reward += float(rew)
          episode_data.append((obs, action, rew, new_obs, done, debug))
          obs = new_obs
          if max_episode_steps is not None and env_step >= \
              max_episode_steps:
            break
        task_step_rewards[task_idx][step_num].append(episode_reward)
        if replay_writer is not None and episode_to_transitions_fn:
          for rec in episode_to_transitions_fn(episode_data):
            replay_writer.write(rec)
        condition_data.append(episode_data)
    avg = np.mean(task_step_rewards[task_idx][
        num_adaptations_per_task - 1])
    _log.info("Task %d avg reward: %f", task_idx, avg)
    if replay_writer is not None and root_dir:
      replay_writer.close()
    if break_after_one_task:
      break

  step_means = []
  for step_num in range(num_adaptations_per_task):
    rewards = [r for t in task_step_rewards.values()
               for r in t[step_num]]
    mean = float(np.mean(rewards)) if rewards else 0.0
    step_means.append(mean)
    if summary_writer is not None:
      summary_writer.add_scalar(f"{tag}/adapt_step_{step_num}_reward",
                                mean, global_step)
  if summary_writer is not None:
    summary_writer.close()
  return step_means
