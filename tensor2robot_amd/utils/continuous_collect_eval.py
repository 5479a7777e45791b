"""Actor process loop: poll for new models, collect + eval episodes.

Reference: `utils/continuous_collect_eval.py:28-108` — loop:
policy.restore() (new ckpt/export) -> run_agent_fn(collect_env) ->
run_agent_fn(eval_env), until global_step >= max_steps; sleeps when no new
model appears.  Any side may die and rejoin (poll-based decoupling).
"""

from __future__ import annotations

import logging
import os
import time
from typing import Callable, Optional

from tensor2robot_amd import gin

_log = logging.getLogger(__name__)


@gin.configurable
def collect_eval_loop(collect_env=None, eval_env=None, policy_class=None,
                      run_agent_fn=None, root_dir: str = "",
                      max_steps: int = 1,
                      poll_sleep_secs: float = 10.0,
                      num_collect_episodes: int = 1,
                      num_eval_episodes: int = 1,
                      max_loops: Optional[int] = None,
                      init_randomly_on_failure: bool = False):
  """The collect/eval actor loop (reference :28-108)."""
  if policy_class is None or run_agent_fn is None:
    raise ValueError("policy_class and run_agent_fn are required")
  policy = policy_class()
  last_global_step = -1
  loops = 0
  while True:
    restored = policy.restore()
    if not restored:
      if init_randomly_on_failure and last_global_step < 0:
        policy.init_randomly()
      else:
        time.sleep(poll_sleep_secs)
        loops += 1
        if max_loops is not None and loops >= max_loops:
          return last_global_step
        continue
    global_step = policy.global_step
    if global_step == last_global_step:
      time.sleep(poll_sleep_secs)  # no new model yet (reference :90-93)
      loops += 1
      if max_loops is not None and loops >= max_loops:
        return last_global_step
      continue
    last_global_step = global_step
    if collect_env is not None:
      # Reference output layout (:76-77): collected episodes land under
      # root_dir/policy_collect, eval episodes under root_dir/eval.
      run_agent_fn(collect_env, policy=policy, global_step=global_step,
                   root_dir=os.path.join(root_dir, "policy_collect")
                   if root_dir else root_dir,
                   num_episodes=num_collect_episodes, tag="collect")
    if eval_env is not None:
      run_agent_fn(eval_env, policy=policy, global_step=global_step,
                   root_dir=os.path.join(root_dir, "eval")
                   if root_dir else root_dir,
                   num_episodes=num_eval_episodes, tag="eval")
    loops += 1
    if global_step >= max_steps:
      return global_step
    if max_loops is not None and loops >= max_loops:
      return last_global_step
