"""Spatial softmax (soft arg-max) feature points.

Reference `layers/spatial_softmax.py:29-89` BuildSpatialSoftmax: reshape
[N,H,W,C] -> [N*C, H*W], softmax (optionally Gumbel-relaxed), expectation
against precomputed x/y position grids in [-1, 1], output [N, 2C] feature
points plus the softmax attention map.

On GPU bf16 channels_last inputs this dispatches to the fused CDNA4
kernel (ops/hip/spatial_softmax.hip): one online-softmax pass per
(image, channel) accumulating max/sum/x/y-expectation in registers plus
one map-write pass, and a single fused backward — replacing the torch
reshape + softmax + [HW,2] matmul chain (5-6 kernels with f32 casts).
The torch composition below remains the CPU/reference path and the
Gumbel-sampling path.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import nn

from tensor2robot_amd import gin
from tensor2robot_amd import ops as ops_mod


class _FusedSpatialSoftmax(torch.autograd.Function):
  """HIP fused soft arg-max; see ops/hip/spatial_softmax.hip."""

  @staticmethod
  def forward(ctx, x, temperature):
    ext = ops_mod.require_hip()
    if not x.is_contiguous(memory_format=torch.channels_last):
      x = x.contiguous(memory_format=torch.channels_last)
    points, smap = ext.spatial_softmax_fwd(x, float(temperature))
    ctx.save_for_backward(smap, points)
    ctx.temperature = float(temperature)
    return points, smap

  @staticmethod
  def backward(ctx, dpoints, dmap):
    ext = ops_mod.require_hip()
    smap, points = ctx.saved_tensors
    dp = None
    if dpoints is not None:
      dp = dpoints.to(torch.bfloat16).contiguous()
    dm = None
    if dmap is not None:
      dm = dmap.to(torch.bfloat16).contiguous(
          memory_format=torch.channels_last)
    dx = ext.spatial_softmax_bwd(smap, points, dp, dm, ctx.temperature)
    return dx, None


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

  def _fused_supported(self, x: torch.Tensor) -> bool:
    import os
    if os.environ.get("T2R_DISABLE_FUSED_SPATIAL_SOFTMAX"):
      return False
    # H*W cap: the kernel loops pixels serially per (image, channel)
    # thread — right for the robot nets' small maps, wrong for huge
    # ones (those fall back to the softmax+GEMM composition).
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4 and
            x.shape[1] <= 256 and x.shape[2] * x.shape[3] <= 4096 and
            not (self.use_gumbel and self.training))

  def forward(self, features: torch.Tensor
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    if self._fused_supported(features):
      return _FusedSpatialSoftmax.apply(features, self.temperature)
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
