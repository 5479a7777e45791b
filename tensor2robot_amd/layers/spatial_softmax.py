"""Spatial softmax (soft arg-max) feature points.

Reference `layers/spatial_softmax.py:29-89` BuildSpatialSoftmax: reshape
[N,H,W,C] -> [N*C, H*W], softmax (optionally Gumbel-relaxed), expectation
against precomputed x/y position grids in [-1, 1], output [N, 2C] feature
points plus the softmax attention map.

Torch-native NCHW.  The softmax + two weighted reductions over H*W is a
single fused pass on GPU bandwidth terms; at the feature-map sizes the
robot nets use it is a minor cost next to the conv tower, so the torch
composition (one softmax + one matmul against the [HW, 2] grid) is kept —
it already fuses the x/y expectations into one GEMM.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import nn

from tensor2robot_amd import gin


def _position_grid(height: int, width: int, device, dtype) -> torch.Tensor:
  """[H*W, 2] grid of (x, y) positions in [-1, 1]."""
  ys = torch.linspace(-1.0, 1.0, height, device=device, dtype=dtype)
  xs = torch.linspace(-1.0, 1.0, width, device=device, dtype=dtype)
  gy, gx = torch.meshgrid(ys, xs, indexing="ij")
  return torch.stack([gx.reshape(-1), gy.reshape(-1)], dim=-1)


@gin.configurable
class SpatialSoftmax(nn.Module):
  """Per-channel soft arg-max; returns ([N, 2C] points, [N,C,H,W] map)."""

  def __init__(self, temperature: float = 1.0,
               use_gumbel: bool = False, gumbel_temperature: float = 1.0):
    super().__init__()
    self.temperature = temperature
    self.use_gumbel = use_gumbel
    self.gumbel_temperature = gumbel_temperature
    self._grid_cache: Optional[Tuple[Tuple[int, int], torch.Tensor]] = None

  def _grid(self, h: int, w: int, device, dtype) -> torch.Tensor:
    key = (h, w)
    if (self._grid_cache is None or self._grid_cache[0] != key or
        self._grid_cache[1].device != device or
        self._grid_cache[1].dtype != dtype):
      self._grid_cache = (key, _position_grid(h, w, device, dtype))
    return self._grid_cache[1]

  def forward(self, features: torch.Tensor
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    n, c, h, w = features.shape
    logits = features.reshape(n * c, h * w).float() / self.temperature
    if self.use_gumbel and self.training:
      # Reference :69-73 RelaxedOneHotCategorical sampling.
      u = torch.rand_like(logits).clamp_(1e-10, 1.0)
      gumbel = -torch.log(-torch.log(u))
      softmax = torch.softmax(
          (logits + gumbel) / self.gumbel_temperature, dim=-1)
    else:
      softmax = torch.softmax(logits, dim=-1)
    grid = self._grid(h, w, features.device, softmax.dtype)
    points = softmax @ grid                       # [N*C, 2]
    # Reference output layout: [N, 2C] = per-channel (x, y) pairs
    # flattened channel-major ([x0 y0 x1 y1 ...] after reshape).
    points = points.reshape(n, c * 2).to(features.dtype)
    return points, softmax.reshape(n, c, h, w).to(features.dtype)


def build_spatial_softmax(features: torch.Tensor, temperature: float = 1.0
                          ) -> Tuple[torch.Tensor, torch.Tensor]:
  """Functional form (reference BuildSpatialSoftmax :29)."""
  return SpatialSoftmax(temperature=temperature)(features)
