"""Gin-configurable schedules of the global step.

Reference `utils/global_step_functions.py`: piecewise_linear — values at
strictly-increasing boundaries with linear interpolation between them,
clamped at the ends.  Returns a python callable(step) -> float so it
plugs into the optimizer factories' `learning_rate` argument.
"""

from __future__ import annotations

from typing import Sequence

from tensor2robot_amd import gin


@gin.configurable
def piecewise_linear(boundaries: Sequence[float],
                     values: Sequence[float]):
  """callable(step) -> interpolated value (reference :26-79)."""
  boundaries = [float(b) for b in boundaries]
  values = [float(v) for v in values]
  assert boundaries and values, "Need non-empty boundaries/values"
  assert len(boundaries) == len(values), \
      "boundaries and values must match"
  assert all(b2 > b1 for b1, b2 in zip(boundaries, boundaries[1:])), \
      "boundaries must be strictly increasing"

  def schedule(step) -> float:
    x = float(step)
    if x <= boundaries[0]:
      return values[0]
    if x >= boundaries[-1]:
      return values[-1]
    for i in range(len(boundaries) - 1):
      if boundaries[i] <= x < boundaries[i + 1]:
        frac = (x - boundaries[i]) / (boundaries[i + 1] - boundaries[i])
        return values[i] + (values[i + 1] - values[i]) * frac
    return values[-1]

  return schedule


@gin.configurable
def exponential_decay(initial_value: float = 0.0001,
                      decay_steps: int = 10000,
                      decay_rate: float = 0.9,
                      staircase: bool = True):
  """callable(step) -> initial_value * decay_rate^(step/decay_steps)
  (reference `global_step_functions.py:98-121`; staircase floors the
  exponent to whole decay intervals like tf.train.exponential_decay)."""
  assert decay_steps > 0, "decay_steps must be positive"

  def schedule(step) -> float:
    p = float(step) / float(decay_steps)
    if staircase:
      p = float(int(p))
    return float(initial_value) * float(decay_rate) ** p

  return schedule
