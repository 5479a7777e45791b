"""MFMAConv2d: hand-written MFMA conv for stride-1 NHWC bf16.

Replaces the reference's TF Conv2D call sites (SURVEY 2.10 item 1:
research/qtopt/networks.py:444-580 5x5/3x3 SAME convs,
film_resnet_model.py:100-105, vision_layers.py:30-160) with CDNA4
MFMA kernels.

Wraps ops/hip/conv_s1.hip. Forward and backward-data both run the MFMA
kernel (backward-data is the same convolution with spatially-flipped,
channel-transposed weights); the weight gradient uses
aten.convolution_backward (MIOpen wrw) — wrw is a [64 x 64] x 200k-K
reduction where MIOpen's split-K igemm is already reasonable.

Weights are prepacked per call to the kernel's LDS-friendly layout
[rs][c/16][k][24] (16 used + 8 pad for conflict-free B-fragment reads)
by one HIP kernel (pack_conv_w); when the input needs a gradient, a
single paired dispatch (pack_conv_w_pair) emits the forward AND the
flipped/transposed dgrad pack together, and the dgrad pack is saved to
backward.  Everything is captured inside hipGraphs along with the conv.
"""

from __future__ import annotations

import os

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import ops as ops_mod

WPAD = 24


def pack_weights(w: torch.Tensor) -> torch.Tensor:
  """[K, C, R, S] -> flat [R*S, C/16, K, 24] bf16 (kernel layout)."""
  k, c, r, s = w.shape
  wp = w.permute(2, 3, 1, 0).reshape(r * s, c // 16, 16, k)
  wp = wp.permute(0, 1, 3, 2).contiguous()          # [rs][c16][k][16]
  out = torch.zeros(r * s, c // 16, k, WPAD, dtype=torch.bfloat16,
                    device=w.device)
  out[..., :16] = wp.to(torch.bfloat16)
  return out


def pack_weights_bwd(w: torch.Tensor) -> torch.Tensor:
  """Pack for backward-data: flip spatially, swap in/out channels."""
  wb = w.flip(2, 3).permute(1, 0, 2, 3).contiguous()
  return pack_weights(wb)


def _supported(x: torch.Tensor, weight: torch.Tensor, stride,
               padding) -> bool:
  if os.environ.get("T2R_DISABLE_MFMA_CONV"):
    return False
  if not (x.is_cuda and x.dtype == torch.bfloat16):
    return False
  k, c, r, s = weight.shape
  # Measured dispatch rule (profiles/): 3.3-4x vs MIOpen on r*s <= 9
  # (all-staged weights, barrier-free loop); 1.07x on the 5x5 via the
  # glds-ring variant.  T2R_MFMA_MAX_RS caps which kernels dispatch
  # here (e.g. 9 = 3x3 only) for A/B against a tuned MIOpen.
  max_rs = int(os.environ.get("T2R_MFMA_MAX_RS", "25"))
  # r == s: the dgrad path derives its pad as (r-1-pad) for BOTH dims
  # and the wrw kernels assume square windows.  c % 32 when the input
  # needs a gradient: dgrad swaps channel roles (output channels = c)
  # and conv_s1_nhwc requires its output-channel count % 32.
  if x.requires_grad and c % 32 != 0:
    return False
  return (stride == (1, 1) and r == s and r * s <= max_rs and
          c % 16 == 0 and c <= 64 and k % 32 == 0 and k <= 64 and
          padding[0] == padding[1])


class _MFMAConvFunction(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, weight, pad):
    ext = ops_mod.require_hip()
    k, c, r, s = weight.shape
    if not x.is_contiguous(memory_format=torch.channels_last):
      x = x.contiguous(memory_format=torch.channels_last)
    if x.requires_grad:
      # One dispatch for both packs; the dgrad pack rides to backward.
      wpk, wpk_b = ext.pack_conv_w_pair(weight)
    else:
      wpk, wpk_b = ext.pack_conv_w(weight, False), None
    y = ext.conv_s1_nhwc(x, wpk, k, r, s, pad)
    if wpk_b is None:
      ctx.save_for_backward(x, weight)
    else:
      ctx.save_for_backward(x, weight, wpk_b)
    ctx.pad = pad
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = ops_mod.require_hip()
    x, weight = ctx.saved_tensors[:2]
    wpk_b = ctx.saved_tensors[2] if len(ctx.saved_tensors) > 2 else None
    k, c, r, s = weight.shape
    dy = dy.contiguous(memory_format=torch.channels_last)
    dx = dw = None
    if ctx.needs_input_grad[0]:
      # SAME-pad duality: the dy->dx conv pad is (R-1-pad).
      bpad = r - 1 - ctx.pad
      if wpk_b is None:
        wpk_b = ext.pack_conv_w(weight, True)
      dx = ext.conv_s1_nhwc(dy.to(torch.bfloat16), wpk_b, c, r, s, bpad)
    if ctx.needs_input_grad[1]:
      # Default: the v4 tr_b16 kernel for the 5x5 (measured 1.05x
      # MIOpen, profiles/); MIOpen for 3x3 (small shapes are launch/
      # staging-bound, v4 0.5x there).  T2R_ENABLE_MFMA_WRW overrides:
      # =1 v1 LDS-accumulator, =2 v2 register-accumulator, =3 v3
      # rs-split, =4 v4 everywhere, =0/off -> MIOpen everywhere.
      wrw_mode = os.environ.get("T2R_ENABLE_MFMA_WRW", "")
      sq35 = c == 64 and k == 64 and r == s and r in (3, 5)
      default_v4 = (wrw_mode == "" and c == 64 and k == 64 and
                    r == 5 and s == 5)
      # (A/B note, gpurun_out/r2b_wrw_ab: on the big-spatial G2V 3x3
      # 64ch @ 118^2, MIOpen wrw steady state is 0.074 ms vs 0.213 for
      # the find-free im2col+GEMM recipe — 2.9x faster.  Its 3.6-s
      # first-call find is one-time and absorbed by the untimed warmup,
      # so MIOpen stays the 3x3 wrw default everywhere; the engine's
      # replay-vs-eager check guards against find-intermediate kernels
      # being baked into a graph.)
      if default_v4 or (wrw_mode == "4" and sq35):
        # v4 emits [K,C,R,S] bf16 directly (permute+cast fused into
        # its reduce kernel).
        dw = ext.conv_s1_wrw4(x, dy.to(torch.bfloat16), r, s, ctx.pad)
        if dw.dtype != weight.dtype:
          dw = dw.to(weight.dtype)
      elif wrw_mode == "3" and sq35:
        dw_f32 = ext.conv_s1_wrw3(x, dy.to(torch.bfloat16), r, s,
                                  ctx.pad)
        dw = dw_f32.reshape(r, s, c, k).permute(3, 2, 0, 1) \
            .contiguous().to(weight.dtype)
      elif wrw_mode == "2" and sq35:
        dw_f32 = ext.conv_s1_wrw2(x, dy.to(torch.bfloat16), r, s,
                                  ctx.pad)
        dw = dw_f32.reshape(r, s, c, k).permute(3, 2, 0, 1) \
            .contiguous().to(weight.dtype)
      elif c % 32 == 0 and wrw_mode == "1":
        # MFMA wrw v1: fp32 LDS-accumulated [rs][c][k] -> [k][c][r][s].
        dw_f32 = ext.conv_s1_wrw(x, dy.to(torch.bfloat16), r, s,
                                 ctx.pad)
        dw = dw_f32.reshape(r, s, c, k).permute(3, 2, 0, 1) \
            .contiguous().to(weight.dtype)
      else:
        # bf16 wrw via MIOpen (an f32 upcast here cost 30% whole-step
        # throughput).
        dw = torch.ops.aten.convolution_backward(
            dy.to(torch.bfloat16), x, weight, None, (1, 1),
            (ctx.pad, ctx.pad), (1, 1), False, (0, 0), 1,
            (False, True, False))[1].to(weight.dtype)
    return dx, dw, None


def _bigc_supported(x: torch.Tensor, weight: torch.Tensor, stride,
                    padding) -> bool:
  """C-chunked MFMA 3x3 path (conv_s1_big.hip): ResNet-family
  C=K in {128,256,512} stride-1 3x3s (film_resnet_model.py:100-341,
  grasp2vec/resnet.py).  Forward and dgrad run the chunk kernel; the
  weight gradient keeps the im2col+GEMM recipe (gemm_conv.dw_from_col).
  Opt-out: T2R_DISABLE_MFMA_BIGC=1 falls back to the GEMM-conv path."""
  if os.environ.get("T2R_DISABLE_MFMA_CONV") or \
      os.environ.get("T2R_DISABLE_MFMA_BIGC"):
    return False
  if not (x.is_cuda and x.dtype == torch.bfloat16):
    return False
  k, c, r, s = weight.shape
  if not (stride == (1, 1) and (r, s) == (3, 3) and
          padding[0] == padding[1] and padding[0] <= 1):
    return False
  # dgrad swaps channel roles, so both must satisfy both constraints
  # (C-role % 32, K-role % 64) when an input gradient is needed.
  if x.requires_grad:
    return c % 64 == 0 and k % 64 == 0 and 96 <= c <= 512 and k <= 512
  return c % 32 == 0 and k % 64 == 0 and 96 <= c <= 512 and k <= 512


class _BigCConvFunction(torch.autograd.Function):
  """Big-channel 3x3: chunked MFMA fwd/dgrad + GEMM weight gradient."""

  @staticmethod
  def forward(ctx, x, weight, pad):
    ext = ops_mod.require_hip()
    k, c, r, s = weight.shape
    if not x.is_contiguous(memory_format=torch.channels_last):
      x = x.contiguous(memory_format=torch.channels_last)
    if x.requires_grad:
      wpk, wpk_b = ext.pack_conv_w_pair(weight)
    else:
      wpk, wpk_b = ext.pack_conv_w(weight, False), None
    y = ext.conv_s1_nhwc_cchunk(x, wpk, k, r, s, pad)
    if wpk_b is None:
      ctx.save_for_backward(x, weight)
    else:
      ctx.save_for_backward(x, weight, wpk_b)
    ctx.pad = pad
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = ops_mod.require_hip()
    x, weight = ctx.saved_tensors[:2]
    wpk_b = ctx.saved_tensors[2] if len(ctx.saved_tensors) > 2 else None
    k, c, r, s = weight.shape
    dy = dy.contiguous(memory_format=torch.channels_last) \
        .to(torch.bfloat16)
    dx = dw = None
    if ctx.needs_input_grad[0]:
      bpad = r - 1 - ctx.pad
      if wpk_b is None:
        wpk_b = ext.pack_conv_w(weight, True)
      dx = ext.conv_s1_nhwc_cchunk(dy, wpk_b, c, r, s, bpad)
    if ctx.needs_input_grad[1]:
      from tensor2robot_amd.ops import gemm_conv
      col = ext.im2col_nhwc(x, r, s, ctx.pad, 1)
      dy_mat = dy.permute(0, 2, 3, 1).reshape(-1, k)
      dw = gemm_conv.dw_from_col(col, dy_mat, r, s, c, k, weight.dtype)
    return dx, dw, None


def _space_to_depth_nhwc(x: torch.Tensor, pad_to: int = 16
                         ) -> torch.Tensor:
  """[N,3,H,W] cl -> [N,pad_to,H/2,W/2] cl, c-order (dr, ds, c)."""
  n, c, h, w = x.shape
  nhwc = x.permute(0, 2, 3, 1)          # view of channels_last
  s2d = nhwc.reshape(n, h // 2, 2, w // 2, 2, c) \
      .permute(0, 1, 3, 2, 4, 5).reshape(n, h // 2, w // 2, 4 * c)
  out = torch.zeros(n, h // 2, w // 2, pad_to, dtype=x.dtype,
                    device=x.device)
  out[..., :4 * c] = s2d
  return out.permute(0, 3, 1, 2).contiguous(
      memory_format=torch.channels_last)


def _stem_weight_s2d(weight: torch.Tensor, pad_to: int = 16
                     ) -> torch.Tensor:
  """[64,3,6,6] -> [64,pad_to,3,3]: w2[k, (dr,ds,c), r2, s2] =
  w[k, c, 2*r2+dr, 2*s2+ds]."""
  k, c, r, s = weight.shape
  w6 = weight.reshape(k, c, 3, 2, 3, 2)         # [k,c,r2,dr,s2,ds]
  w2 = w6.permute(0, 3, 5, 1, 2, 4).reshape(k, 4 * c, 3, 3)
  out = torch.zeros(k, pad_to, 3, 3, dtype=weight.dtype,
                    device=weight.device)
  out[:, :4 * c] = w2
  return out


class _StemConvFunction(torch.autograd.Function):
  """6x6/2 C=3 stem via space-to-depth + the fast 3x3/1 MFMA path.

  The 2x2-block reshape turns the strided 6x6 into an exactly
  equivalent 3x3 stride-1 conv with C=12 (zero-padded to 16) — the
  STAGE_ALL conv_s1 kernel's sweet spot.  The stem is the network's
  first layer, so no input gradient is produced.
  """

  @staticmethod
  def forward(ctx, x, weight):
    ext = ops_mod.require_hip()
    if not x.is_contiguous(memory_format=torch.channels_last):
      x = x.contiguous(memory_format=torch.channels_last)
    x_s2d = ext.s2d_stem(x)
    w_s2d = _stem_weight_s2d(weight)
    wpk = ext.pack_conv_w(w_s2d, False)
    y = ext.conv_s1_nhwc(x_s2d, wpk, weight.shape[0], 3, 3, 1)
    ctx.save_for_backward(x, weight)
    return y

  @staticmethod
  def backward(ctx, dy):
    x, weight = ctx.saved_tensors
    dy = dy.contiguous(memory_format=torch.channels_last)
    dw = None
    if ctx.needs_input_grad[1]:
      dw = torch.ops.aten.convolution_backward(
          dy.to(torch.bfloat16), x, weight, None, (2, 2), (2, 2),
          (1, 1), False, (0, 0), 1,
          (False, True, False))[1].to(weight.dtype)
    return None, dw


def _stem_supported(x, weight, stride, padding) -> bool:
  # Space-to-depth route measured 1.13x MIOpen end-to-end (profiles/).
  if os.environ.get("T2R_DISABLE_MFMA_CONV") or \
      os.environ.get("T2R_DISABLE_MFMA_STEM"):
    return False
  if not (x.is_cuda and x.dtype == torch.bfloat16):
    return False
  k, c, r, s = weight.shape
  return (stride == (2, 2) and padding == (2, 2) and (k, c, r, s) ==
          (64, 3, 6, 6) and not x.requires_grad)


class MFMAConv2d(nn.Conv2d):
  """Conv2d that runs the MFMA kernel on supported GPU bf16 shapes."""

  def __init__(self, in_channels, out_channels, kernel_size, stride=1,
               padding=0, bias=False):
    super().__init__(in_channels, out_channels, kernel_size,
                     stride=stride, padding=padding, bias=bias)

  def forward(self, x):
    if x.is_cuda and x.dtype == torch.float32 and \
        torch.is_autocast_enabled():
      # Autocast would cast inside F.conv2d anyway; casting HERE lets
      # the dispatch checks see the bf16 tensor (the f32 preprocess
      # output otherwise sent the 6x6 stem to MIOpen - profiles/).
      x = x.to(torch.get_autocast_dtype("cuda"))
    if self.bias is None and _supported(x, self.weight, self.stride,
                                        self.padding):
      w = self.weight
      if w.dtype != torch.bfloat16:
        w = w.to(torch.bfloat16)
      return _MFMAConvFunction.apply(x, w, self.padding[0])
    if self.bias is None and _bigc_supported(x, self.weight, self.stride,
                                             self.padding):
      w = self.weight
      if w.dtype != torch.bfloat16:
        w = w.to(torch.bfloat16)
      return _BigCConvFunction.apply(x, w, self.padding[0])
    if self.bias is None and _stem_supported(x, self.weight, self.stride,
                                             self.padding):
      w = self.weight
      if w.dtype != torch.bfloat16:
        w = w.to(torch.bfloat16)
      return _StemConvFunction.apply(x, w)
    if self.bias is None:
      # ResNet-family shapes (C/K up to 512, stride 1/2, 1x1/3x3):
      # hand im2col/col2im + rocBLAS GEMM (ops/gemm_conv.py) instead of
      # MIOpen's helper-kernel storm on these shapes.
      from tensor2robot_amd.ops import gemm_conv
      if gemm_conv.supported(x, self.weight, self.stride, self.padding,
                             self.dilation, self.groups):
        return gemm_conv.gemm_conv2d(x, self.weight, self.stride,
                                     self.padding)
      if gemm_conv.supported_cpad(x, self.weight, self.stride,
                                  self.padding, self.dilation,
                                  self.groups):
        return gemm_conv.gemm_conv2d_cpad(x, self.weight, self.stride,
                                          self.padding)
    if x.is_cuda and os.environ.get("T2R_LOG_CONV_FALLBACK"):
      print(f"# conv fallback: x={tuple(x.shape)} {x.dtype} "
            f"req_grad={x.requires_grad} w={tuple(self.weight.shape)} "
            f"stride={self.stride} pad={self.padding}", flush=True)
    if x.is_cuda and x.dtype != self.weight.dtype and \
        not torch.is_autocast_enabled():
      # bf16 activations outside autocast: run the fallback in bf16 too.
      return F.conv2d(x, self.weight.to(x.dtype), None, self.stride,
                      self.padding, self.dilation, self.groups)
    return super().forward(x)
