"""Mock model + data for hermetic integration tests.

Reference `utils/mocks.py`: MockT2RModel (:99, 3-layer FC + BN over a
3-float feature), MockInputGenerator (:43, deterministic linearly-separable
numpy dataset), MockExportGenerator (:33).  Integration tests train to
near-convergence in seconds and assert real learning.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.data import input_generators
from tensor2robot_amd.export_generators import abstract_export_generator
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.specs import tensorspec_utils as tsu

TSPEC = tsu.ExtendedTensorSpec


class MockNetwork(torch.nn.Module):

  def __init__(self, input_dim: int = 3, hidden: int = 16):
    super().__init__()
    self.stack = torch.nn.Sequential(
        torch.nn.Linear(input_dim, hidden),
        torch.nn.BatchNorm1d(hidden),
        torch.nn.ReLU(),
        torch.nn.Linear(hidden, hidden),
        torch.nn.ReLU(),
        torch.nn.Linear(hidden, 1),
    )

  def forward(self, x):
    return self.stack(x)


@gin.configurable
class MockT2RModel(abstract_model.AbstractT2RModel):
  """Tiny binary classifier over a 3-float feature (reference mocks:99)."""

  def __init__(self, multi_dataset: bool = False, device_type: str = "cpu",
               **kwargs):
    kwargs.setdefault("compute_dtype", "float32")
    super().__init__(device_type=device_type, **kwargs)
    self._multi_dataset = multi_dataset

  def get_feature_specification(self, mode):
    spec = tsu.TensorSpecStruct()
    if self._multi_dataset:
      spec["x1"] = TSPEC(shape=(3,), dtype=torch.float32, name="measured_position",
                         dataset_key="dataset1")
      spec["x2"] = TSPEC(shape=(3,), dtype=torch.float32, name="measured_position",
                         dataset_key="dataset2")
    else:
      spec["measured_position"] = TSPEC(shape=(3,), dtype=torch.float32,
                                        name="measured_position")
    return spec

  def get_label_specification(self, mode):
    spec = tsu.TensorSpecStruct()
    if self._multi_dataset:
      spec["valid_position"] = TSPEC(shape=(1,), dtype=torch.float32,
                                     name="valid_position",
                                     dataset_key="dataset1")
    else:
      spec["valid_position"] = TSPEC(shape=(1,), dtype=torch.float32,
                                     name="valid_position")
    return spec

  def create_network(self):
    return MockNetwork()

  def inference_network_fn(self, features, labels, mode, params=None):
    if self._multi_dataset:
      x = features.x1
    else:
      x = features.measured_position
    logit = self.network(x)
    return {"logit": logit, "prediction": torch.sigmoid(logit)}

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    loss = torch.nn.functional.binary_cross_entropy_with_logits(
        inference_outputs["logit"], labels.valid_position)
    return loss

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    pred = (inference_outputs["prediction"] > 0.5).float()
    accuracy = (pred == labels.valid_position).float().mean()
    return {"accuracy": accuracy}

  def create_export_outputs_fn(self, features, inference_outputs, mode,
                               params=None):
    return {"logit": inference_outputs["logit"],
            "prediction": inference_outputs["prediction"]}


@gin.configurable
class MockInputGenerator(input_generators.GeneratorInputGenerator):
  """Deterministic linearly separable data (reference mocks:43-71).

  Label is 1 when sum(x) > 0, with a margin for fast convergence.
  """

  def __init__(self, batch_size: int = 8, seed: int = 7, **kwargs):
    super().__init__(batch_size=batch_size, **kwargs)
    self._rng = np.random.RandomState(seed)

  def _generate_batch(self, batch_index):
    x = self._rng.uniform(-1.0, 1.0, size=(self._batch_size, 3)).astype(
        np.float32)
    margin = np.sum(x, axis=1)
    y = (margin > 0).astype(np.float32)[:, None]
    # Push points away from the decision boundary for separability.
    x = x + 0.3 * np.sign(margin)[:, None]
    features = tsu.TensorSpecStruct()
    features["measured_position"] = x.astype(np.float32)
    labels = tsu.TensorSpecStruct()
    labels["valid_position"] = y
    return features, labels


class MockExportGenerator(
    abstract_export_generator.AbstractExportGenerator):
  pass


class MockTF2T2RModel(MockT2RModel):
  """Reference mocks.py MockTF2T2RModel: the TF2-flavored mock is the
  same torch model here (no session/graph split to emulate)."""
