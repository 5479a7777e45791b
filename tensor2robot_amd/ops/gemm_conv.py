"""GEMM-conv: the ResNet-family convolution path on MI355X.

For the ResNet/FiLM shapes (C or K up to 512, stride 1/2, 1x1 and 3x3;
reference film_resnet_model.py:100-341, research/bcz/model.py:245-288,
grasp2vec/resnet.py), MIOpen on this pool dispatches thousands of
SubTensorOp/fillBuffer helper kernels per step and parks in find
(profiles/ r2: BC-Z FiLM-ResNet18 measured 21 ms/step).  The
MI355X-first shape for these convs is ONE hand gather kernel + ONE
library GEMM per direction — rocBLAS bf16 GEMMs run at near-peak MFMA
and im2col/col2im are clean 16-B-per-lane streams (ops/hip/im2col.hip):

  fwd : y = im2col(x) @ w_mat                        (1x1 skips im2col)
  dw  : dw_mat = col^T @ dy_mat
  dx  : dx = col2im(dy_mat @ w_mat^T)                (1x1: GEMM only)

The QT-Opt C=K=64 stride-1 shapes keep the fused MFMA kernels in
conv_s1.hip (halo reuse beats im2col's R*S read amplification there);
this module covers what those kernels do not.
"""

from __future__ import annotations

import os as _os

import torch

_DBG = bool(_os.environ.get("T2R_GEMM_DEBUG"))

from tensor2robot_amd import ops as ops_mod


def _w_mat(weight: torch.Tensor) -> torch.Tensor:
  """[K, C, R, S] -> [R*S*C, K] bf16, (r, s, c) row order = col order."""
  k = weight.shape[0]
  return weight.permute(2, 3, 1, 0).reshape(-1, k).to(torch.bfloat16) \
      .contiguous()


def dw_from_col(col, dy_mat, r, s, c, k, out_dtype) -> torch.Tensor:
  """dw = col^T @ dy_mat -> [K,C,R,S]; chunked for huge contractions.

  hipBLASLt runs tiny-output x huge-contraction GEMMs without split-K
  and strands the chip (measured 56 TF at M=668k vs 177 TF chunked —
  tools/probe_gemm_shapes.py); chunked bmm + sum restores grid
  parallelism.  Shared by the GEMM-conv path and the big-C MFMA conv's
  weight gradient.
  """
  m = col.shape[0]
  if m >= 65536:
    chunks = max(2, min(64, m // 8192))
    mc = m // chunks
    head = chunks * mc
    dw_mat = torch.bmm(
        col[:head].view(chunks, mc, -1).transpose(1, 2),
        dy_mat[:head].view(chunks, mc, k)).sum(0)
    if head < m:
      dw_mat = dw_mat + col[head:].t() @ dy_mat[head:]
  else:
    dw_mat = col.t() @ dy_mat                    # [RS*C, K]
  return dw_mat.reshape(r, s, c, k).permute(3, 2, 0, 1) \
      .contiguous().to(out_dtype)


class _GemmConvFunction(torch.autograd.Function):
  """im2col + rocBLAS GEMM conv (NHWC bf16, stride 1/2)."""

  @staticmethod
  def forward(ctx, x, weight, stride, pad):
    ext = ops_mod.require_hip()
    if not x.is_contiguous(memory_format=torch.channels_last):
      if _DBG:
        print(f"# gemm fwd x copy {tuple(x.shape)} strides={x.stride()}",
              flush=True)
      x = x.contiguous(memory_format=torch.channels_last)
    n, c, h, w = x.shape
    k, _, r, s = weight.shape
    oh = (h + 2 * pad - r) // stride + 1
    ow = (w + 2 * pad - s) // stride + 1
    wm = _w_mat(weight)
    one_by_one = (r == 1 and s == 1 and stride == 1 and pad == 0)
    if one_by_one:
      col = x.permute(0, 2, 3, 1).reshape(-1, c)  # NHWC view, no copy
    else:
      col = ext.im2col_nhwc(x, r, s, pad, stride)
    y = torch.empty((n, k, oh, ow), dtype=torch.bfloat16,
                    device=x.device
                    ).contiguous(memory_format=torch.channels_last)
    # channels_last [N,K,OH,OW] storage IS [M,K] row-major: GEMM
    # writes straight into it, no copy.
    torch.matmul(col, wm, out=y.permute(0, 2, 3, 1).reshape(-1, k))
    # Save x, NOT col: col is R*S*x bytes (multi-GB on big-spatial
    # ResNet towers) and is a cheap gather to recompute in backward.
    ctx.save_for_backward(x, weight, wm)
    ctx.conf = (stride, pad, one_by_one)
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = ops_mod.require_hip()
    x, weight, wm = ctx.saved_tensors
    stride, pad, one_by_one = ctx.conf
    n, c, h, w = x.shape
    k, _, r, s = weight.shape
    if _DBG and not dy.is_contiguous(memory_format=torch.channels_last):
      print(f"# gemm bwd dy copy {tuple(dy.shape)} strides={dy.stride()}",
            flush=True)
    dy = dy.contiguous(memory_format=torch.channels_last) \
        .to(torch.bfloat16)
    dy_mat = dy.permute(0, 2, 3, 1).reshape(-1, k)
    dx = dw = None
    if one_by_one:
      col = x.permute(0, 2, 3, 1).reshape(-1, c)
    else:
      col = ext.im2col_nhwc(x, r, s, pad, stride)
    if ctx.needs_input_grad[1]:
      dw = dw_from_col(col, dy_mat, r, s, c, k, weight.dtype)
    if ctx.needs_input_grad[0]:
      if one_by_one:
        # GEMM writes straight into the cl dx storage — the earlier
        # dcol-then-copy_ paid a full extra dx pass per 1x1 conv
        # (one of the top elementwise slices in the G2V profile).
        dx = torch.empty_like(x)
        torch.matmul(dy_mat, wm.t(),
                     out=dx.permute(0, 2, 3, 1).reshape(-1, c))
      else:
        dcol = dy_mat @ wm.t()                     # [M, RS*C]
        dx = ext.col2im_nhwc(dcol, n, c, h, w, r, s, pad, stride)
    return dx, dw, None, None


def supported(x: torch.Tensor, weight: torch.Tensor, stride, padding,
              dilation, groups) -> bool:
  import os
  if os.environ.get("T2R_DISABLE_GEMM_CONV"):
    return False
  if not (x.is_cuda and x.dtype == torch.bfloat16):
    return False
  if groups != 1 or dilation != (1, 1):
    return False
  if stride[0] != stride[1] or stride[0] not in (1, 2):
    return False
  if padding[0] != padding[1]:
    return False
  k, c, r, s = weight.shape
  if c % 8 != 0 or k % 8 != 0:
    return False
  if r > 7 or s > 7:
    return False
  return True


def gemm_conv2d(x, weight, stride, padding) -> torch.Tensor:
  w = weight if weight.dtype == torch.bfloat16 else \
      weight.to(torch.bfloat16)
  return _GemmConvFunction.apply(x, w, stride[0], padding[0])


def supported_cpad(x, weight, stride, padding, dilation, groups) -> bool:
  """Small-C stems (RGB 7x7/2 etc.): zero-pad channels to 8 and run the
  GEMM path — MIOpen's wrw find on these burns minutes (profiles/)."""
  import os
  if os.environ.get("T2R_DISABLE_GEMM_CONV"):
    return False
  if not (x.is_cuda and x.dtype == torch.bfloat16):
    return False
  if groups != 1 or dilation != (1, 1):
    return False
  if stride[0] != stride[1] or stride[0] not in (1, 2):
    return False
  if padding[0] != padding[1]:
    return False
  k, c, r, s = weight.shape
  return 0 < c < 8 and k % 8 == 0 and r <= 7 and s <= 7


def gemm_conv2d_cpad(x, weight, stride, padding) -> torch.Tensor:
  n, c, h, w = x.shape
  pad_c = 8 - c
  zx = x.new_zeros((n, pad_c, h, w))
  xp = torch.cat([x, zx], 1).contiguous(
      memory_format=torch.channels_last)
  wzero = weight.new_zeros((weight.shape[0], pad_c, weight.shape[2],
                            weight.shape[3]))
  wp = torch.cat([weight, wzero], 1).to(torch.bfloat16)
  return _GemmConvFunction.apply(xp, wp, stride[0], padding[0])
