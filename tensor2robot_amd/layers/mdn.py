"""Mixture density network heads.

Reference `layers/mdn.py`: `get_mixture_distribution` :30 (params
[..., A + 2*A*S] -> Categorical(alphas) x MultivariateNormalDiag(mus,
softplus(sigmas) + 1e-4) :67-72), `predict_mdn_params` :76 (FC head with
optionally unconditioned sigmas as free variables :104-113),
`gaussian_mixture_approximate_mode` :117 (mean of the most probable
component), `MDNDecoder` :128 (stateful decoder: __call__ -> action,
loss(labels) = mean NLL :164-167).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin

SIGMA_EPS = 1e-4  # reference :70


class GaussianMixture:
  """Diag-gaussian mixture over the last axis; batch dims arbitrary.

  Lightweight native replacement for tfp's Categorical x MVNDiag mixture:
  only what T2R uses — log_prob, sample, approximate mode, mean.
  """

  def __init__(self, logits: torch.Tensor, mus: torch.Tensor,
               sigmas: torch.Tensor):
    # logits [..., A]; mus/sigmas [..., A, S]
    self.logits = logits
    self.mus = mus
    self.sigmas = sigmas

  def log_prob(self, x: torch.Tensor) -> torch.Tensor:
    """x [..., S] -> [...] log p(x)."""
    x = x.unsqueeze(-2)                             # [..., 1, S]
    var = self.sigmas ** 2
    comp_lp = -0.5 * (((x - self.mus) ** 2) / var
                      + 2.0 * torch.log(self.sigmas)
                      + math.log(2.0 * math.pi)).sum(dim=-1)  # [..., A]
    mix_lp = torch.log_softmax(self.logits, dim=-1)
    return torch.logsumexp(mix_lp + comp_lp, dim=-1)

  def sample(self, generator: Optional[torch.Generator] = None
             ) -> torch.Tensor:
    probs = torch.softmax(self.logits, dim=-1)
    flat = probs.reshape(-1, probs.shape[-1])
    idx = torch.multinomial(flat, 1, generator=generator).reshape(
        probs.shape[:-1] + (1, 1))
    idx = idx.expand(probs.shape[:-1] + (1, self.mus.shape[-1]))
    mu = torch.gather(self.mus, -2, idx).squeeze(-2)
    sigma = torch.gather(self.sigmas, -2, idx).squeeze(-2)
    noise = torch.randn(mu.shape, generator=generator, device=mu.device,
                        dtype=mu.dtype)
    return mu + sigma * noise

  def approximate_mode(self) -> torch.Tensor:
    """Mean of the most probable component (reference :117-124)."""
    idx = torch.argmax(self.logits, dim=-1, keepdim=True).unsqueeze(-1)
    idx = idx.expand(self.logits.shape[:-1] + (1, self.mus.shape[-1]))
    return torch.gather(self.mus, -2, idx).squeeze(-2)

  def mean(self) -> torch.Tensor:
    w = torch.softmax(self.logits, dim=-1).unsqueeze(-1)
    return (w * self.mus).sum(dim=-2)


def get_mixture_distribution(params: torch.Tensor, num_alphas: int,
                             sample_size: int,
                             output_mean: Optional[torch.Tensor] = None
                             ) -> GaussianMixture:
  """params [..., A + 2AS] -> mixture (reference :30-73)."""
  a, s = num_alphas, sample_size
  if params.shape[-1] != a + 2 * a * s:
    raise ValueError(
        f"params last dim {params.shape[-1]} != {a + 2 * a * s}")
  logits = params[..., :a]
  mus = params[..., a: a + a * s].reshape(params.shape[:-1] + (a, s))
  sig_raw = params[..., a + a * s:].reshape(params.shape[:-1] + (a, s))
  sigmas = F.softplus(sig_raw) + SIGMA_EPS
  if output_mean is not None:
    # Broadcast a [S]-shaped (or [..., S]) mean across components.
    mus = mus + output_mean.unsqueeze(-2)
  return GaussianMixture(logits, mus, sigmas)


def gaussian_mixture_approximate_mode(mixture: GaussianMixture
                                      ) -> torch.Tensor:
  return mixture.approximate_mode()


@gin.configurable
class MDNHead(nn.Module):
  """FC head producing mixture params (reference predict_mdn_params :76).

  condition_sigmas=False keeps sigmas as free learned variables
  independent of the input (reference :104-113).
  """

  def __init__(self, in_dim: int, sample_size: int, num_alphas: int = 5,
               condition_sigmas: bool = False):
    super().__init__()
    self.num_alphas = num_alphas
    self.sample_size = sample_size
    self.condition_sigmas = condition_sigmas
    a, s = num_alphas, sample_size
    out = a + a * s + (a * s if condition_sigmas else 0)
    self.fc = nn.Linear(in_dim, out)
    if not condition_sigmas:
      self.sigma_param = nn.Parameter(torch.zeros(a * s))

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    """Returns params [..., A + 2AS] in get_mixture_distribution layout."""
    out = self.fc(x)
    if not self.condition_sigmas:
      sig = self.sigma_param.expand(out.shape[:-1] + self.sigma_param.shape)
      out = torch.cat([out, sig], dim=-1)
    return out

  def distribution(self, x: torch.Tensor) -> GaussianMixture:
    return get_mixture_distribution(self(x), self.num_alphas,
                                    self.sample_size)


@gin.configurable
class MDNDecoder(nn.Module):
  """Stateful decoder: forward -> action, loss(labels) -> NLL.

  Reference `mdn.py:128-167`: __call__ runs the head, stores the mixture,
  returns the approximate mode as the action; loss(labels) is the mean
  negative log-likelihood of the stored mixture.
  """

  def __init__(self, in_dim: int, action_size: int, num_mixture: int = 5,
               condition_sigmas: bool = False):
    super().__init__()
    self.head = MDNHead(in_dim, action_size, num_alphas=num_mixture,
                        condition_sigmas=condition_sigmas)
    self._mixture: Optional[GaussianMixture] = None
    self._params: Optional[torch.Tensor] = None

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    self._params = self.head(x)
    self._mixture = get_mixture_distribution(
        self._params, self.head.num_alphas, self.head.sample_size)
    return self._mixture.approximate_mode()

  def loss(self, labels: torch.Tensor) -> torch.Tensor:
    if self._mixture is None:
      raise RuntimeError("MDNDecoder.loss called before forward")
    from tensor2robot_amd.ops import mdn_nll as fused
    if fused.supported(self._params, self.head.num_alphas):
      # One HIP kernel per direction instead of the eager op chain
      # (ops/hip/mdn_nll.hip); identical math incl. softplus + 1e-4.
      return fused.mdn_nll(self._params, labels,
                           self.head.num_alphas,
                           self.head.sample_size).mean()
    return -self._mixture.log_prob(labels).mean()
