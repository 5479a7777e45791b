"""Pose toy environment: CPU end-to-end smoke workload.

Re-design of the reference's PyBullet duck-pose env
(`research/pose_env/pose_env.py:51-120`) without a physics/render
dependency: a synthetic 64x64 render of a blob at a 2D pose; the agent
predicts the pose; reward = negative distance.  `hidden_drift` offsets the
target pose per-task for meta-learning (reference :84-89).
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from tensor2robot_amd import gin

IMAGE_SIZE = 64


def render_pose_image(pose: np.ndarray, size: int = IMAGE_SIZE
                      ) -> np.ndarray:
  """Renders an RGB uint8 image with a gaussian blob at `pose` in [-1,1]^2."""
  y = (pose[1] * 0.5 + 0.5) * (size - 1)
  x = (pose[0] * 0.5 + 0.5) * (size - 1)
  yy, xx = np.meshgrid(np.arange(size), np.arange(size), indexing="ij")
  d2 = (yy - y) ** 2 + (xx - x) ** 2
  blob = np.exp(-d2 / (2.0 * 4.0 ** 2))
  img = np.zeros((size, size, 3), np.float32)
  img[:, :, 0] = blob          # red blob
  img[:, :, 1] = 0.2 * blob
  img[:, :, 2] = 1.0 - blob    # blue background
  return (img * 255).astype(np.uint8)


@gin.configurable
class PoseToyEnv:
  """Single-step pose-regression episodes."""

  def __init__(self, hidden_drift: bool = False,
               drift_scale: float = 0.3, seed: Optional[int] = None):
    self._rng = np.random.RandomState(seed)
    self._hidden_drift = hidden_drift
    self._drift_scale = drift_scale
    self._drift = np.zeros(2, np.float32)
    self._pose = np.zeros(2, np.float32)
    if hidden_drift:
      self.reset_task()

  def reset_task(self):
    """New hidden drift: the meta-learning task variable (reference :84)."""
    self._drift = self._rng.uniform(
        -self._drift_scale, self._drift_scale, 2).astype(np.float32)

  def reset(self) -> np.ndarray:
    self._pose = self._rng.uniform(-0.8, 0.8, 2).astype(np.float32)
    return render_pose_image(self._pose)

  @property
  def target_pose(self) -> np.ndarray:
    return self._pose + self._drift

  def step(self, action: np.ndarray):
    action = np.asarray(action, np.float32).reshape(-1)[:2]
    dist = float(np.linalg.norm(action - self.target_pose))
    reward = -dist
    obs = render_pose_image(self._pose)
    return obs, reward, True, {"distance": dist,
                               "target_pose": self.target_pose.copy()}


@gin.configurable
class PoseEnvRandomPolicy:
  """Uniform-random pose guesses (reference :35-48)."""

  def __init__(self, seed: Optional[int] = None):
    self._rng = np.random.RandomState(seed)

  def reset(self):
    pass

  def restore(self):
    return True

  def init_randomly(self):
    pass

  @property
  def global_step(self):
    return 0

  def sample_action(self, obs, explore_prob: float = 0.0):
    del obs, explore_prob
    return self._rng.uniform(-1, 1, 2).astype(np.float32), {}

  def SelectAction(self, state, context=None, timestep: int = 0):
    return self._rng.uniform(-1, 1, 2).astype(np.float32)


def get_pybullet_urdf_root() -> str:
  """Reference pose_env.py:25-31 resolves pybullet_data's URDF path;
  this rebuild replaces PyBullet with the synthetic blob env, so the
  path is only meaningful when pybullet happens to be installed."""
  try:
    import pybullet_data  # type: ignore
    return pybullet_data.getDataPath()
  except ImportError as e:
    raise ImportError(
        "pybullet is not part of this MI355X image; the synthetic "
        "PoseEnv (this module) replaces it") from e
