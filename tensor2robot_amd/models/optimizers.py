"""Optimizer factories + EMA (moving-average) machinery.

Reference: `models/optimizers.py` (gin factories :27-131, moving-average
optimizer + swapping saver :133-159) and `research/qtopt/optimizer_builder.py`.

Factories return a callable `params -> torch.optim.Optimizer` so models can
declare `create_optimizer_fn` via gin.  The EMA wrapper keeps shadow
parameters updated after every step; checkpoints store the AVERAGED weights
(swapping-saver semantics, `abstract_model.py:855-863`): `swap_in/swap_out`
exchange live and averaged weights around eval/export/checkpoint.
"""

from __future__ import annotations

import math
from typing import Callable, Dict, Iterable, Optional

import torch

from tensor2robot_amd import gin


@gin.configurable
def create_constant_learning_rate(initial_learning_rate: float = 1e-4):
  """lr schedule: constant (reference :27-31)."""
  return lambda step: initial_learning_rate


@gin.configurable
def create_exp_decaying_learning_rate(initial_learning_rate: float = 1e-4,
                                      decay_steps: int = 10000,
                                      decay_rate: float = 0.9,
                                      staircase: bool = True):
  """lr schedule: exponential decay (reference :33-60)."""

  def schedule(step: int) -> float:
    p = step / float(decay_steps)
    if staircase:
      p = math.floor(p)
    return initial_learning_rate * (decay_rate ** p)

  return schedule


def _resolve_lr(learning_rate):
  if callable(learning_rate):
    return learning_rate
  return lambda step: float(learning_rate)


class ScheduledOptimizer:
  """Wraps torch.optim with a per-step lr schedule + grad clipping."""

  def __init__(self, optimizer: torch.optim.Optimizer,
               lr_schedule: Callable[[int], float],
               clip_gradient_norm: float = 0.0):
    self.optimizer = optimizer
    self.lr_schedule = lr_schedule
    self.clip_gradient_norm = clip_gradient_norm

  def step(self, global_step: int):
    lr = self.lr_schedule(global_step)
    for group in self.optimizer.param_groups:
      group["lr"] = lr
    if self.clip_gradient_norm > 0:
      params = [p for g in self.optimizer.param_groups for p in g["params"]]
      torch.nn.utils.clip_grad_norm_(params, self.clip_gradient_norm)
    self.optimizer.step()

  def zero_grad(self, set_to_none: bool = True):
    self.optimizer.zero_grad(set_to_none=set_to_none)

  def state_dict(self):
    return self.optimizer.state_dict()

  def load_state_dict(self, state):
    self.optimizer.load_state_dict(state)

  @property
  def param_groups(self):
    return self.optimizer.param_groups


@gin.configurable
def default_create_optimizer_fn(learning_rate=1e-4,
                                clip_gradient_norm: float = 0.0):
  """Adam with default lr (reference :62-67)."""
  return create_adam_optimizer(learning_rate,
                               clip_gradient_norm=clip_gradient_norm)


@gin.configurable
def create_adam_optimizer(learning_rate=1e-4, beta1: float = 0.9,
                          beta2: float = 0.999, epsilon: float = 1e-8,
                          clip_gradient_norm: float = 0.0,
                          fused: bool = True):
  """Adam factory (reference :69-81)."""
  schedule = _resolve_lr(learning_rate)

  def build(params) -> ScheduledOptimizer:
    use_cuda = torch.cuda.is_available()
    # capturable: Adam's step counter lives on-device so the optimizer
    # step can be recorded into a hipGraph (the Trainer's fast path).
    opt = torch.optim.Adam(params, lr=schedule(0), betas=(beta1, beta2),
                           eps=epsilon, fused=fused and use_cuda,
                           capturable=use_cuda)
    return ScheduledOptimizer(opt, schedule, clip_gradient_norm)

  return build


@gin.configurable
def create_sgd_optimizer(learning_rate=1e-4, clip_gradient_norm: float = 0.0):
  """SGD factory (reference :83-106)."""
  schedule = _resolve_lr(learning_rate)

  def build(params) -> ScheduledOptimizer:
    opt = torch.optim.SGD(params, lr=schedule(0))
    return ScheduledOptimizer(opt, schedule, clip_gradient_norm)

  return build


@gin.configurable
def create_momentum_optimizer(learning_rate=1e-4, momentum: float = 0.9,
                              use_nesterov: bool = False,
                              clip_gradient_norm: float = 0.0):
  """Momentum factory (reference :108-131; the QT-Opt default)."""
  schedule = _resolve_lr(learning_rate)

  def build(params) -> ScheduledOptimizer:
    opt = torch.optim.SGD(params, lr=schedule(0), momentum=momentum,
                          nesterov=use_nesterov)
    return ScheduledOptimizer(opt, schedule, clip_gradient_norm)

  return build


@gin.configurable
def create_rms_prop_optimizer(learning_rate=1e-4, decay: float = 0.9,
                              momentum: float = 0.0, epsilon: float = 1e-10,
                              clip_gradient_norm: float = 0.0):
  schedule = _resolve_lr(learning_rate)

  def build(params) -> ScheduledOptimizer:
    opt = torch.optim.RMSprop(params, lr=schedule(0), alpha=decay,
                              momentum=momentum, eps=epsilon,
                              capturable=torch.cuda.is_available())
    return ScheduledOptimizer(opt, schedule, clip_gradient_norm)

  return build


class ExponentialMovingAverage:
  """EMA of model parameters with swap-in/out (swapping-saver semantics).

  The update runs as two batched multi-tensor kernels
  (torch._foreach_mul_/add_ — one fused launch each over every shadow
  tensor) and captures inside hipGraphs; this class is the
  orchestration around them.
  """

  def __init__(self, module: torch.nn.Module, decay: float = 0.9999):
    self.decay = decay
    self._module = module
    self.shadow: Dict[str, torch.Tensor] = {
        name: p.detach().clone()
        for name, p in module.named_parameters() if p.requires_grad
    }
    self._backup: Optional[Dict[str, torch.Tensor]] = None

  @torch.no_grad()
  def update(self):
    d = self.decay
    names = list(self.shadow.keys())
    params = dict(self._module.named_parameters())
    shadows = [self.shadow[n] for n in names]
    currents = [params[n].detach() for n in names]
    # shadow = d * shadow + (1-d) * param, batched.
    torch._foreach_mul_(shadows, d)
    torch._foreach_add_(shadows, currents, alpha=1.0 - d)

  @torch.no_grad()
  def swap_in(self):
    """Puts averaged weights into the live module (for eval/export/ckpt)."""
    if self._backup is not None:
      raise RuntimeError("EMA already swapped in")
    params = dict(self._module.named_parameters())
    self._backup = {n: params[n].detach().clone() for n in self.shadow}
    for n, s in self.shadow.items():
      params[n].copy_(s)

  @torch.no_grad()
  def swap_out(self):
    if self._backup is None:
      raise RuntimeError("EMA not swapped in")
    params = dict(self._module.named_parameters())
    for n, b in self._backup.items():
      params[n].copy_(b)
    self._backup = None

  def state_dict(self):
    return {"decay": self.decay, "shadow": self.shadow}

  def load_state_dict(self, state):
    self.decay = state["decay"]
    for k, v in state["shadow"].items():
      if k in self.shadow:
        self.shadow[k].copy_(v)
      else:
        self.shadow[k] = v.clone()


@gin.configurable
def create_moving_average_optimizer(decay: float = 0.9999):
  """Returns the EMA decay config used by the train loop (reference :133)."""
  return decay


@gin.configurable
def create_gradient_descent_optimizer(
    learning_rate=1e-4, clip_gradient_norm: float = 0.0):
  """Plain gradient descent factory (reference optimizers.py:83-100) —
  SGD without momentum."""
  return create_sgd_optimizer(learning_rate=learning_rate,
                              clip_gradient_norm=clip_gradient_norm)


@gin.configurable
def create_swapping_saver(ema, model_dir: str,
                          keep_checkpoint_every_n_hours: float = 1.0,
                          max_to_keep: int = 5):
  """Saver that writes the EMA (averaged) weights (reference
  optimizers.py:150-160 swapping_saver): every save swaps the averaged
  parameters in around the snapshot — the MovingAverageOptimizer
  swapping-saver contract, bound to our Checkpointer."""
  from tensor2robot_amd.train import checkpointing
  ckpt = checkpointing.Checkpointer(
      model_dir, max_to_keep=max_to_keep,
      keep_checkpoint_every_n_hours=keep_checkpoint_every_n_hours)

  class _SwappingSaver:
    checkpointer = ckpt

    def save(self, step, network, optimizer=None, extra=None):
      return ckpt.save(step, network, optimizer=optimizer, ema=ema,
                       extra=extra)

    def __getattr__(self, name):
      return getattr(ckpt, name)

  return _SwappingSaver()
