"""Fused MDN negative log-likelihood (ops/hip/mdn_nll.hip).

One kernel per direction replaces the ~12-op eager chain torch builds
for `GaussianMixture.log_prob` + its backward (layers/mdn.py; reference
layers/mdn.py:67-72,164-167).  `mdn_nll(params, labels, A, S)` returns
per-row NLL [M]; take `.mean()` for the decoder loss.  Row layout is
get_mixture_distribution's: [logits(A) | mus(AS) | raw_sigmas(AS)],
sigma = softplus(raw) + 1e-4.
"""

from __future__ import annotations

import torch

from tensor2robot_amd import ops as ops_mod


class _MDNNLLFunction(torch.autograd.Function):

  @staticmethod
  def forward(ctx, params, labels, a, s):
    ext = ops_mod.require_hip()
    nll, wsave = ext.mdn_nll_forward(params, labels, a, s)
    ctx.save_for_backward(params, labels, wsave)
    ctx.dims = (a, s)
    return nll.view(params.shape[:-1])

  @staticmethod
  def backward(ctx, gout):
    ext = ops_mod.require_hip()
    params, labels, wsave = ctx.saved_tensors
    a, s = ctx.dims
    dparams = ext.mdn_nll_backward(params, labels, wsave,
                                   gout.reshape(-1), a, s)
    return dparams.view(params.shape), None, None, None


def supported(params: torch.Tensor, num_alphas: int) -> bool:
  import os
  if os.environ.get("T2R_DISABLE_FUSED_MDN"):
    return False
  return (params.is_cuda and num_alphas <= 32 and
          params.dtype in (torch.float32, torch.bfloat16))


def mdn_nll(params: torch.Tensor, labels: torch.Tensor, num_alphas: int,
            sample_size: int) -> torch.Tensor:
  """params [..., A+2AS], labels [..., S] -> NLL [...] (graph-capturable)."""
  return _MDNNLLFunction.apply(params.contiguous(), labels, num_alphas,
                               sample_size)
