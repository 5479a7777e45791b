"""Default MSE action decoder.

Reference `research/vrgripper/mse_decoder.py:26-36`: a linear layer from
params to actions; loss is MSE against labels.action.
"""

from __future__ import annotations

import torch
from torch import nn

from tensor2robot_amd import gin


@gin.configurable
class MSEDecoder(nn.Module):
  """Linear head + MSE loss (reference :26)."""

  def __init__(self, in_dim: int, output_size: int):
    super().__init__()
    self.fc = nn.Linear(in_dim, output_size)
    self._predictions = None

  def forward(self, params: torch.Tensor) -> torch.Tensor:
    self._predictions = self.fc(params)
    return self._predictions

  def loss(self, labels) -> torch.Tensor:
    return torch.nn.functional.mse_loss(self._predictions,
                                        labels["action"])
