"""Cross-entropy method (CEM) optimizer for action selection.

Reference: `utils/cross_entropy.py:30-...` — iterative sample -> evaluate ->
elite-refit loop over Gaussian action distributions; supports dict or array
sample batches and early termination.
"""

from __future__ import annotations

from typing import Callable, Optional, Tuple

import numpy as np

from tensor2robot_amd import gin


@gin.configurable
class CrossEntropyMethod:

  def __init__(self, num_samples: int = 64, num_elites: int = 6,
               num_iterations: int = 3,
               early_termination_value: Optional[float] = None,
               seed: Optional[int] = None):
    self.num_samples = num_samples
    self.num_elites = num_elites
    self.num_iterations = num_iterations
    self.early_termination_value = early_termination_value
    self._rng = np.random.RandomState(seed)

  def run(self, objective_fn: Callable[[np.ndarray], np.ndarray],
          initial_mean: np.ndarray, initial_std: np.ndarray,
          bounds: Optional[Tuple[np.ndarray, np.ndarray]] = None
          ) -> Tuple[np.ndarray, float, np.ndarray, np.ndarray]:
    """Maximizes objective_fn over actions.

    objective_fn: [num_samples, action_dim] -> [num_samples] scores.
    Returns (best_action, best_score, final_mean, final_std).
    """
    mean = np.asarray(initial_mean, np.float32).copy()
    std = np.asarray(initial_std, np.float32).copy()
    best_action, best_score = mean.copy(), -np.inf
    for _ in range(self.num_iterations):
      samples = self._rng.normal(
          mean[None, :], std[None, :],
          size=(self.num_samples, mean.shape[0])).astype(np.float32)
      if bounds is not None:
        samples = np.clip(samples, bounds[0], bounds[1])
      scores = np.asarray(objective_fn(samples)).reshape(-1)
      order = np.argsort(-scores)
      elites = samples[order[: self.num_elites]]
      if scores[order[0]] > best_score:
        best_score = float(scores[order[0]])
        best_action = samples[order[0]].copy()
      mean = elites.mean(axis=0)
      std = elites.std(axis=0) + 1e-6
      if self.early_termination_value is not None and \
          best_score >= self.early_termination_value:
        break
    return best_action, best_score, mean, std


@gin.configurable
def cross_entropy_optimize(objective_fn, action_size: int,
                           num_samples: int = 64, num_elites: int = 6,
                           num_iterations: int = 3,
                           initial_std: float = 0.5,
                           seed: Optional[int] = None):
  """Functional convenience wrapper."""
  cem = CrossEntropyMethod(num_samples=num_samples, num_elites=num_elites,
                           num_iterations=num_iterations, seed=seed)
  mean = np.zeros(action_size, np.float32)
  std = np.full(action_size, initial_std, np.float32)
  return cem.run(objective_fn, mean, std)


def cross_entropy_method(sample_fn, objective_fn, update_fn,
                         initial_params, num_elites: int,
                         num_iterations: int = 1,
                         threshold_to_terminate: Optional[float] = None):
  """Generic CEM maximization (reference `cross_entropy.py:30-107`).

  Sample batches may be lists `[x0..xn]` or dicts of such lists; the
  elite selection sorts ascending by value and keeps the top
  `num_elites`, exactly as the reference.  Returns
  (final_samples, final_values, final_params).
  """
  import operator
  updated_params = initial_params
  samples, values = None, None
  for _ in range(num_iterations):
    samples = sample_fn(**updated_params)
    values = list(objective_fn(samples))
    if isinstance(samples, dict):
      order = [i for i, _ in sorted(enumerate(values),
                                    key=operator.itemgetter(1))]
      elite_samples = {
          k: [v[i] for i in order][-num_elites:]
          for k, v in samples.items()}
    else:
      sorted_samples = [
          s for s, _ in sorted(zip(samples, values),
                               key=operator.itemgetter(1))]
      elite_samples = sorted_samples[-num_elites:]
    updated_params = update_fn(updated_params, elite_samples)
    if (threshold_to_terminate is not None and
        max(values) > threshold_to_terminate):
      break
  return samples, values, updated_params


def normal_cross_entropy_method(objective_fn, mean, stddev,
                                num_samples: int, num_elites: int,
                                num_iterations: int = 1):
  """CEM with a normal sampler (reference `cross_entropy.py:110-156`).

  Returns (final_mean, final_stddev); the elite refit uses Bessel's
  correction (ddof=1) like the reference.
  """
  size = np.broadcast(mean, stddev).size

  def _sample_fn(mean, stddev):
    return np.asarray(mean) + np.asarray(stddev) * np.random.randn(
        num_samples, size)

  def _update_fn(params, elite_samples):
    del params
    elite = np.asarray(elite_samples)
    return {"mean": np.mean(elite, axis=0),
            "stddev": np.std(elite, axis=0, ddof=1)}

  _, _, final_params = cross_entropy_method(
      _sample_fn, objective_fn, _update_fn,
      {"mean": mean, "stddev": stddev}, num_elites,
      num_iterations=num_iterations)
  return final_params["mean"], final_params["stddev"]


# Reference-exact names (utils/cross_entropy.py:30,110).
CrossEntropyMethodFn = cross_entropy_method
NormalCrossEntropyMethod = normal_cross_entropy_method
