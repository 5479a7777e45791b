"""pose_env episode -> serialized training Examples.

Reference `research/pose_env/episode_to_transitions.py:32-...`:
episode_to_transitions_pose_toy serializes (image, action-pose, reward,
target_pose) per transition — the supervised-regression replay format
that the pose_env regression model's parser consumes.
"""

from __future__ import annotations

from typing import List

import numpy as np

from tensor2robot_amd import gin
from tensor2robot_amd.data import example as example_mod
from tensor2robot_amd.data import image_codec


@gin.configurable
def episode_to_transitions_pose_toy(episode_data) -> List[bytes]:
  """One serialized Example per transition (reference :32)."""
  transitions = []
  for obs_t, action, reward, _obs_tp1, _done, debug in episode_data:
    features = {
        "state/image": [image_codec.encode_png(
            np.asarray(obs_t, np.uint8))],
        "pose": np.asarray(action, np.float32).reshape(-1),
        "reward": np.asarray([reward], np.float32),
    }
    if debug and "target_pose" in debug:
      features["target_pose"] = np.asarray(debug["target_pose"],
                                           np.float32).reshape(-1)
    transitions.append(example_mod.encode_example(features))
  return transitions
