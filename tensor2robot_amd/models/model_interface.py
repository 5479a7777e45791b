"""Infrastructure-facing model contract (reference models/model_interface.py:47)."""

from __future__ import annotations

import abc


class ModelInterface(abc.ABC):
  """Minimal contract the train/eval/export infrastructure relies on."""

  @abc.abstractmethod
  def get_feature_specification(self, mode):
    pass

  @abc.abstractmethod
  def get_label_specification(self, mode):
    pass

  def get_feature_specification_for_packing(self, mode):
    return self.preprocessor.get_out_feature_specification(mode)

  def get_label_specification_for_packing(self, mode):
    return self.preprocessor.get_out_label_specification(mode)

  @property
  @abc.abstractmethod
  def preprocessor(self):
    pass

  @property
  @abc.abstractmethod
  def device_type(self) -> str:
    pass

  @property
  def is_device_gpu(self) -> bool:
    return self.device_type == "gpu"

  @property
  def is_device_cpu(self) -> bool:
    return self.device_type == "cpu"

  @abc.abstractmethod
  def model_fn(self, features, labels, mode, params=None):
    pass
