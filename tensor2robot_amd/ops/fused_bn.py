"""FusedBatchNormReLU: drop-in BN(+ReLU) module backed by HIP kernels.

On CUDA (ROCm) bf16 channels-last tensors this runs the hand-written CDNA4
kernels (ops/hip/fused_bn_relu.hip): 2 forward + 3 backward kernels instead
of MIOpen's BN chain + separate ReLU (see profiles/ for the baseline cost).
On CPU it runs the plain torch reference (also used by numerics tests).
"""

from __future__ import annotations

import torch
from torch import nn

from tensor2robot_amd import ops as ops_mod


class _FusedBNReLUFunction(torch.autograd.Function):
  """4D-native: the autograd boundary carries the channels_last NCHW
  tensor itself.  The earlier flat-[M,C]-view boundary made autograd
  materialize a contiguous grad copy per BN layer per step (~26 bf16
  copy kernels/step in the flagship profile); here the kernels read the
  cl storage through zero-copy views in both directions."""

  @staticmethod
  def forward(ctx, x, gamma, beta, running_mean, running_var, eps,
              momentum, fuse_relu):
    ext = ops_mod.require_hip()
    shape_info = None
    x_flat = x
    if x.dim() == 4:
      n, c, h, w = x.shape
      shape_info = (n, c, h, w)
      x_flat = x.permute(0, 2, 3, 1).reshape(n * h * w, c)  # view of cl
    y_flat, stats = ext.fused_bn_relu_forward(
        x_flat, gamma, beta, running_mean, running_var, eps, momentum,
        fuse_relu)
    ctx.save_for_backward(x_flat, gamma, beta, stats)
    ctx.fuse_relu = fuse_relu
    ctx.shape_info = shape_info
    if shape_info is None:
      return y_flat
    return y_flat.view(n, h, w, c).permute(0, 3, 1, 2)

  @staticmethod
  def backward(ctx, dy):
    ext = ops_mod.require_hip()
    x_flat, gamma, beta, stats = ctx.saved_tensors
    shape_info = ctx.shape_info
    if shape_info is not None:
      n, c, h, w = shape_info
      if not dy.is_contiguous(memory_format=torch.channels_last):
        dy = dy.contiguous(memory_format=torch.channels_last)
      dy_flat = dy.permute(0, 2, 3, 1).reshape(n * h * w, c)  # view
    else:
      dy_flat = dy.contiguous()
    dx_flat, grads = ext.fused_bn_relu_backward(
        dy_flat, x_flat, gamma, beta, stats, ctx.fuse_relu)
    if shape_info is None:
      dx = dx_flat
    else:
      dx = dx_flat.view(n, h, w, c).permute(0, 3, 1, 2)
    return dx, grads[1], grads[0], None, None, None, None, None


def _flat_nhwc(x: torch.Tensor):
  """[N,C,H,W] channels_last or [N,C] -> flat [M,C] view + restore info."""
  if x.dim() == 2:
    return x.contiguous(), None
  if x.dim() == 4:
    if not x.is_contiguous(memory_format=torch.channels_last):
      x = x.contiguous(memory_format=torch.channels_last)
    n, c, h, w = x.shape
    flat = x.permute(0, 2, 3, 1).reshape(n * h * w, c)
    return flat, (n, c, h, w)
  raise ValueError(f"FusedBatchNormReLU supports 2D/4D, got {x.dim()}D")


def _unflat(y_flat: torch.Tensor, shape_info):
  if shape_info is None:
    return y_flat
  n, c, h, w = shape_info
  return y_flat.reshape(n, h, w, c).permute(0, 3, 1, 2).contiguous(
      memory_format=torch.channels_last)


class FusedBatchNormReLU(nn.Module):
  """BatchNorm2d/1d + optional ReLU; HIP-fused on GPU bf16."""

  def __init__(self, num_features: int, eps: float = 1e-3,
               momentum: float = 0.003, fuse_relu: bool = True):
    super().__init__()
    self.num_features = num_features
    self.eps = eps
    self.momentum = momentum
    self.fuse_relu = fuse_relu
    self.weight = nn.Parameter(torch.ones(num_features))
    self.bias = nn.Parameter(torch.zeros(num_features))
    self.register_buffer("running_mean", torch.zeros(num_features))
    self.register_buffer("running_var", torch.ones(num_features))
    self.register_buffer("num_batches_tracked",
                         torch.tensor(0, dtype=torch.long))
    # Python-side mirror: the per-forward GPU increment was a captured
    # kernel per BN layer per step (~2% of the train step); the count
    # is only bookkeeping (momentum is explicit), so it lives on the
    # host and is synced into the buffer when a state_dict is taken.
    self._batches_tracked_py = 0

  def _use_hip(self, x: torch.Tensor) -> bool:
    return (x.is_cuda and x.dtype == torch.bfloat16 and
            self.num_features % 8 == 0 and self.num_features <= 2048)

  def forward(self, x: torch.Tensor,
              relu: "bool | None" = None) -> torch.Tensor:
    """Run BN(+ReLU).  `relu` overrides the constructor's fuse_relu for
    this call — FiLM sites need BN -> film -> ReLU (reference
    film_resnet_model.py:210-212,333-335), so the block splits the
    fusion only when a gamma_beta is actually present."""
    fuse_relu = self.fuse_relu if relu is None else relu
    if self._use_hip(x):
      if self.training:
        if x.dim() == 4 and not x.is_contiguous(
            memory_format=torch.channels_last):
          x = x.contiguous(memory_format=torch.channels_last)
        elif x.dim() == 2:
          x = x.contiguous()
        y = _FusedBNReLUFunction.apply(
            x, self.weight, self.bias,
            self.running_mean, self.running_var, self.eps, self.momentum,
            fuse_relu)
        self._batches_tracked_py += 1
        return y
      flat, shape_info = _flat_nhwc(x)
      invstd = torch.rsqrt(self.running_var + self.eps)
      scale = (self.weight * invstd).float()
      shift = (self.bias - self.running_mean * self.weight * invstd
               ).float()
      y = ops_mod.require_hip().bn_inference_apply(
          flat, scale.contiguous(), shift.contiguous(), fuse_relu)
      return _unflat(y, shape_info)
    # Torch reference path (CPU / non-bf16): identical math.
    y = torch.nn.functional.batch_norm(
        x, self.running_mean, self.running_var, self.weight, self.bias,
        self.training, self.momentum, self.eps)
    if fuse_relu:
      y = torch.relu(y)
    return y

  def _save_to_state_dict(self, destination, prefix, keep_vars):
    if self._batches_tracked_py:
      self.num_batches_tracked.fill_(
          int(self.num_batches_tracked) + self._batches_tracked_py)
      self._batches_tracked_py = 0
    super()._save_to_state_dict(destination, prefix, keep_vars)

  def extra_repr(self):
    return (f"{self.num_features}, eps={self.eps}, "
            f"momentum={self.momentum}, fuse_relu={self.fuse_relu}")
