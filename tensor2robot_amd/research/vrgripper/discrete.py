"""Discrete (binned) action decoder.

Reference `research/vrgripper/discrete.py`: GetDiscreteBins :29,
GetDiscreteActions :48 (argmax over per-dim bin softmax, dotted with bin
centers), GetDiscreteActionLoss :78 (nearest-bin one-hot + softmax CE),
DiscreteDecoder :107.
"""

from __future__ import annotations

from typing import Optional, Sequence

import numpy as np
import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin


def get_discrete_bins(num_bins: int, output_min: np.ndarray,
                      output_max: np.ndarray) -> np.ndarray:
  """[num_bins, action_dim] bin centers (reference :29-45)."""
  action_range = output_max - output_min
  bin_sizes = action_range / float(num_bins)
  return np.array([output_min + bin_sizes * (b + 0.5)
                   for b in range(num_bins)])


def get_discrete_actions(logits: torch.Tensor, action_size: int,
                         num_bins: int,
                         bin_centers: np.ndarray) -> torch.Tensor:
  """Mode action from per-dim bin logits (reference :48-76)."""
  probs = torch.softmax(logits.reshape(-1, action_size, num_bins), dim=-1)
  onehot = F.one_hot(probs.argmax(-1), num_bins).to(probs.dtype)
  centers = torch.as_tensor(bin_centers.T, dtype=probs.dtype,
                            device=logits.device)  # [action_dim, bins]
  actions = (onehot * centers).sum(-1)
  return actions.reshape(*logits.shape[:-1], action_size)


def get_discrete_action_loss(logits: torch.Tensor,
                             action_labels: torch.Tensor,
                             bin_centers: np.ndarray,
                             num_bins: int) -> torch.Tensor:
  """Nearest-bin CE loss (reference :78-104)."""
  centers = torch.as_tensor(bin_centers, dtype=action_labels.dtype,
                            device=action_labels.device)  # [bins, dim]
  labels = action_labels.unsqueeze(-2)  # [..., 1, dim]
  while centers.dim() < labels.dim():
    centers = centers.unsqueeze(0)
  discrete = ((labels - centers) ** 2).argmin(-2)  # [..., dim]
  onehot = F.one_hot(discrete, num_bins).float().reshape(-1, num_bins)
  flat_logits = logits.reshape(-1, num_bins)
  ce = -(onehot * torch.log_softmax(flat_logits, dim=-1)).sum(-1)
  return ce.mean()


@gin.configurable
class DiscreteDecoder(nn.Module):
  """Per-dim binned classification decoder (reference :107-137)."""

  def __init__(self, in_dim: int, output_size: int, num_bins: int = 1,
               output_min: Optional[Sequence[float]] = None,
               output_max: Optional[Sequence[float]] = None):
    super().__init__()
    self._num_bins = num_bins
    self._output_size = output_size
    self._bin_centers = get_discrete_bins(
        num_bins, np.asarray(output_min, np.float32),
        np.asarray(output_max, np.float32))
    self.head = nn.Linear(in_dim, output_size * num_bins)
    self._action_logits = None

  def forward(self, params: torch.Tensor) -> torch.Tensor:
    self._action_logits = self.head(params)
    return get_discrete_actions(self._action_logits, self._output_size,
                                self._num_bins, self._bin_centers)

  def loss(self, labels) -> torch.Tensor:
    return get_discrete_action_loss(self._action_logits,
                                    labels["action"],
                                    self._bin_centers, self._num_bins)
