"""FusedMaxPool2d: non-overlapping NHWC bf16 max-pool on HIP.

Drop-in for nn.MaxPool2d(k, stride=k, ceil_mode=...) in the grasping
nets (reference slim max_pool2d call sites, qtopt/networks.py:460,534;
SURVEY 2.10 item 4) via ops/hip/maxpool.hip: forward stores a window-local uint8 argmax;
backward is a conflict-free gather — torch's atomic max_pool_backward
was 7.4% of the QT-Opt steady-state step (profiles/).

CPU / non-bf16 inputs fall back to F.max_pool2d (same math; numerics
tests compare the two).
"""

from __future__ import annotations

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import ops as ops_mod


class _FusedMaxPoolFunction(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, kh, kw, ceil_mode):
    ext = ops_mod.require_hip()
    y, argmax = ext.maxpool_nhwc_forward(x, kh, kw, ceil_mode)
    ctx.save_for_backward(argmax)
    ctx.in_shape = x.shape
    ctx.kh, ctx.kw = kh, kw
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = ops_mod.require_hip()
    (argmax,) = ctx.saved_tensors
    n, c, h, w = ctx.in_shape
    dx = ext.maxpool_nhwc_backward(dy, argmax, n, c, h, w, ctx.kh,
                                   ctx.kw)
    return dx, None, None, None


class FusedMaxPool2d(nn.Module):
  """Max-pool with stride == kernel (non-overlapping windows)."""

  def __init__(self, kernel_size: int, stride=None,
               ceil_mode: bool = True):
    super().__init__()
    stride = stride if stride is not None else kernel_size
    if stride != kernel_size:
      raise ValueError("FusedMaxPool2d requires stride == kernel_size "
                       f"(got k={kernel_size}, s={stride})")
    self.kernel_size = kernel_size
    self.ceil_mode = ceil_mode

  def _use_hip(self, x: torch.Tensor) -> bool:
    import os
    if os.environ.get("T2R_DISABLE_FUSED_MAXPOOL"):
      return False
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4 and
            x.shape[1] % 8 == 0)

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    if self._use_hip(x):
      if not x.is_contiguous(memory_format=torch.channels_last):
        x = x.contiguous(memory_format=torch.channels_last)
      return _FusedMaxPoolFunction.apply(x, self.kernel_size,
                                         self.kernel_size,
                                         self.ceil_mode)
    return F.max_pool2d(x, self.kernel_size, stride=self.kernel_size,
                        ceil_mode=self.ceil_mode)
