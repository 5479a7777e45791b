"""Episode -> serialized transition records for replay writers.

Reference `research/vrgripper/episode_to_transitions.py`:
make_fixed_length :40 (temporal resampling keeping endpoints),
episode_to_transitions_reacher :82 (per-transition tf.Example),
episode_to_transitions_metareacher :104 (one SequenceExample per
episode).  Records use the native proto codec (data/example.py).
"""

from __future__ import annotations

import collections
from typing import List, Optional

import numpy as np

from tensor2robot_amd import gin
from tensor2robot_amd.data import example as example_mod


@gin.configurable
def make_fixed_length(input_list, fixed_length: int,
                      always_include_endpoints: bool = True,
                      randomized: bool = True,
                      rng: Optional[np.random.RandomState] = None):
  """Samples entries to a fixed length (reference :40-77)."""
  original_length = len(input_list)
  if original_length <= 2:
    return None
  if not randomized:
    indices = np.sort(np.mod(np.arange(fixed_length), original_length))
    return [input_list[i] for i in indices]
  rng = rng or np.random
  if always_include_endpoints:
    endpoint_indices = np.array([0, original_length - 1])
    other_indices = 1 + rng.choice(original_length - 2, fixed_length - 2,
                                   replace=True)
    indices = np.concatenate((endpoint_indices, other_indices), axis=0)
  else:
    indices = rng.choice(original_length, fixed_length, replace=True)
  indices = np.sort(indices)
  return [input_list[i] for i in indices]


def _floats(v):
  return np.asarray(v, np.float32).reshape(-1)


@gin.configurable
def episode_to_transitions_reacher(episode_data, is_demo: bool = False
                                   ) -> List[bytes]:
  """One serialized Example per transition (reference :82-101)."""
  transitions = []
  for obs_t, action, reward, obs_tp1, done, _ in episode_data:
    features = {
        "pose_t": _floats(obs_t),
        "pose_tp1": _floats(obs_tp1),
        "action": _floats(action),
        "reward": _floats([reward]),
        "done": np.asarray([int(done)], np.int64),
        "is_demo": np.asarray([int(is_demo)], np.int64),
    }
    transitions.append(example_mod.encode_example(features))
  return transitions


@gin.configurable
def episode_to_transitions_metareacher(episode_data) -> List[bytes]:
  """One serialized SequenceExample per episode (reference :104-131)."""
  context = {
      "is_demo": np.asarray([int(episode_data[0][-1]["is_demo"])],
                            np.int64),
      "target_idx": np.asarray([episode_data[0][-1]["target_idx"]],
                               np.int64),
  }
  feature_lists = collections.defaultdict(list)
  for obs_t, action, reward, obs_tp1, done, _ in episode_data:
    feature_lists["pose_t"].append(_floats(obs_t))
    feature_lists["pose_tp1"].append(_floats(obs_tp1))
    feature_lists["action"].append(_floats(action))
    feature_lists["reward"].append(_floats([reward]))
    feature_lists["done"].append(np.asarray([int(done)], np.int64))
  return [example_mod.encode_sequence_example(context,
                                              dict(feature_lists))]
