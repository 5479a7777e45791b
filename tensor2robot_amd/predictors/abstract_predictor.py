"""Predictor contracts: inference-time model loading on numpy I/O.

Reference: `predictors/abstract_predictor.py:26-81`.
"""

from __future__ import annotations

import abc
from typing import Dict, Optional

import numpy as np


class AbstractPredictor(abc.ABC):

  @abc.abstractmethod
  def predict(self, features: Dict[str, np.ndarray]
              ) -> Dict[str, np.ndarray]:
    """Runs inference on a {flat_key: ndarray} feed."""

  @abc.abstractmethod
  def get_feature_specification(self):
    pass

  def get_label_specification(self):
    return None

  @abc.abstractmethod
  def restore(self) -> bool:
    """(Re)loads the newest model artifacts; returns success."""

  def init_randomly(self):
    raise NotImplementedError(
        f"{type(self).__name__} does not support random init")

  def close(self):
    pass

  def assert_is_loaded(self):
    if not self.is_loaded:
      raise ValueError(f"{type(self).__name__} has no model loaded; call "
                       "restore() or init_randomly() first.")

  @property
  @abc.abstractmethod
  def is_loaded(self) -> bool:
    pass

  @property
  def model_version(self) -> int:
    return self.global_step

  @property
  @abc.abstractmethod
  def global_step(self) -> int:
    pass

  @property
  @abc.abstractmethod
  def model_path(self) -> Optional[str]:
    pass

  def __enter__(self):
    return self

  def __exit__(self, *exc):
    self.close()
