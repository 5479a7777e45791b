"""FastStepEngine: the hipGraph-captured, flat-grad-view training step.

This is the framework home of the performance-engineered step that
round 1 kept inside bench.py (bench.py:164-313 then): the Trainer uses
it for every GPU training run, and bench.py is a thin caller.

Design (MI355X-first, replaces the reference's in-runtime step
execution — `tpu_model_wrapper.py:45-49,236` CrossShardOptimizer and
the Estimator train loop `utils/train_eval.py:424-613`):

  * Single GPU: ONE hipGraph holds zero_grad + forward + backward +
    optimizer + EMA — a ~450-kernel step becomes one graph launch.
  * Distributed (one process per MI355X, RCCL over xGMI): a flat f32
    comm buffer with every `p.grad` pre-assigned as a strided VIEW into
    it (channels_last strides for 4D params), so backward ACCUMULATES
    straight into the comm buffer.  graph1 = zero+fwd+bwd, then ONE
    eager `all_reduce(flat)` per step (the only eager op — fewer,
    larger collectives suit the 7-link point-to-point xGMI fabric),
    then graph2 = div + optimizer + EMA.
  * The capture-or-eager decision is COLLECTIVE (all_reduce MIN): one
    rank replaying graphs while another runs the hook-bucketed eager
    engine would mismatch collectives and hang the job.
  * Learning-rate schedules: a capture freezes the python-side lr, so
    the engine re-captures the optimizer graph when the schedule moves
    by >0.1% relative; if that happens too often (continuous decay) it
    demotes itself to the eager path.

Dynamic work — on-GPU preprocessing with host-side RNG — stays OUTSIDE
the graph: callers copy preprocessed batches into the engine's static
input buffers each step (`step(features, labels)` does this).
"""

from __future__ import annotations

import logging
import time
from typing import Callable, Dict, Optional, Tuple

import torch
import torch.distributed as dist

from tensor2robot_amd.parallel import graph_step
from tensor2robot_amd.specs import tensorspec_utils as tsu

_log = logging.getLogger(__name__)

# Set while an engine is settling/benchmarking/capturing: background
# producers (the pinned-ring pipeline) stay off the GPU runtime so
# MIOpen/hipBLASLt algorithm timing is clean.
import threading
CAPTURE_QUIESCE = threading.Event()


def _flatten_tensors(struct) -> Dict[str, torch.Tensor]:
  """TensorSpecStruct/dict of tensors -> flat {path: tensor}."""
  if struct is None:
    return {}
  out = {}
  for key, value in struct.items():
    if isinstance(value, torch.Tensor):
      out[key] = value
  return out


def _clone_static(t: torch.Tensor) -> torch.Tensor:
  if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last):
    return torch.empty_like(t, memory_format=torch.channels_last)
  return torch.empty_like(t)


class FastStepEngine:
  """Graph-captured train step with distributed gradient communication.

  Protocol:
    engine = FastStepEngine(network, optimizer, ema, device, ...)
    engine.build(loss_fn, example_features, example_labels)
    loss = engine.step(features, labels)     # per training step

  `loss_fn(features, labels)` must return the scalar loss reading only
  the passed structs + module state (e.g. a model_fn wrapper).  The
  example batch defines the static buffer shapes; every later batch
  must match (drop_remainder batching guarantees this).
  """

  def __init__(self, network: torch.nn.Module,
               optimizer, ema=None,
               device: Optional[torch.device] = None,
               use_graph: bool = True,
               autocast_dtype: torch.dtype = torch.bfloat16,
               lr_schedule: Optional[Callable[[int], float]] = None,
               max_recaptures: int = 8):
    self.network = network
    self.optimizer = optimizer
    self.ema = ema
    self.device = device or next(network.parameters()).device
    self.use_cuda = self.device.type == "cuda"
    self.distributed = dist.is_available() and dist.is_initialized()
    self.world_size = dist.get_world_size() if self.distributed else 1
    self._want_graph = use_graph and self.use_cuda
    self._autocast_dtype = autocast_dtype
    self._lr_schedule = lr_schedule
    self._captured_lr = None
    self._recaptures = 0
    self._max_recaptures = max_recaptures
    self.graphed = None
    self.opt_graphed = None
    self._static_f: Dict[str, torch.Tensor] = {}
    self._static_l: Dict[str, torch.Tensor] = {}
    self._flat = None
    self._loss_fn = None
    self._built = False
    self._dp_engine = None  # eager fallback engine

  # -- helpers ---------------------------------------------------------------
  def _autocast(self):
    return torch.autocast(device_type="cuda", dtype=self._autocast_dtype,
                          enabled=self.use_cuda)

  def _grad_params(self):
    return [p for p in self.network.parameters() if p.requires_grad]

  def _static_structs(self) -> Tuple:
    features = tsu.TensorSpecStruct()
    for key, t in self._static_f.items():
      features[key] = t
    labels = None
    if self._static_l:
      labels = tsu.TensorSpecStruct()
      for key, t in self._static_l.items():
        labels[key] = t
    return features, labels

  def _assign_grad_views(self):
    """Pre-assign every p.grad as a strided view into the flat buffer."""
    off = 0
    for p in self._grad_params():
      n = p.numel()
      sl = self._flat[off:off + n]
      if p.dim() == 4 and p.is_contiguous(
          memory_format=torch.channels_last):
        no, c, h, w = p.shape
        g = sl.view(no, h, w, c).permute(0, 3, 1, 2)
      else:
        g = sl.view(p.shape)
      p.grad = g
      off += n

  def _check_grad_aliasing(self):
    """Backward must ACCUMULATE into the views; a rebound p.grad would
    make the all-reduce sync a dead buffer."""
    base = self._flat.data_ptr()
    end = base + self._flat.numel() * self._flat.element_size()
    for p in self._grad_params():
      if not (base <= p.grad.data_ptr() < end):
        raise RuntimeError("grad view rebound during capture")

  # -- build -----------------------------------------------------------------
  def build(self, loss_fn: Callable, example_features, example_labels,
            settle_steps: int = 3, global_step: int = 0):
    """Allocates statics, settles, captures.  Collective across ranks."""
    self._loss_fn = loss_fn
    for key, t in _flatten_tensors(example_features).items():
      self._static_f[key] = _clone_static(t)
    for key, t in _flatten_tensors(example_labels).items():
      self._static_l[key] = _clone_static(t)
    self._copy_in(example_features, example_labels)

    if self.distributed:
      # Identical start on every rank (the reference's chief-initialized
      # variables semantics).
      for p in self.network.parameters():
        dist.broadcast(p.data, src=0)

    ok = True
    if self._want_graph:
      CAPTURE_QUIESCE.set()
      try:
        self._capture(global_step)
      except Exception as e:  # pragma: no cover - runtime-dependent
        _log.warning("hipGraph capture unavailable, eager fallback: %r", e)
        self.graphed = None
        self.opt_graphed = None
        ok = False
      finally:
        CAPTURE_QUIESCE.clear()
    else:
      ok = False

    if self.distributed:
      # Collective demotion: everyone graphs, or nobody does.
      flag = torch.tensor([1.0 if ok else 0.0],
                          device=self.device if self.use_cuda else "cpu")
      dist.all_reduce(flag, op=dist.ReduceOp.MIN)
      if float(flag.item()) < 1.0:
        self._demote()
      # The settle steps before capture ran unsynced; re-align ranks
      # (in-place writes — captured graphs read these tensors).
      for p in self.network.parameters():
        dist.broadcast(p.data, src=0)
    elif not ok:
      self._demote()
    self._built = True

  def _capture(self, global_step: int, attempt: int = 0):
    static_features, static_labels = self._static_structs()
    for i in range(3):  # settle MIOpen algo find before capture
      self._eager_step_inner(static_features, static_labels,
                             global_step)
    torch.cuda.synchronize()
    # Baseline for the post-capture sanity check: a capture taken while
    # MIOpen's find is mid-flight bakes find-intermediate (slow)
    # kernels into the graph permanently — measured 13 ms/step vs 2.8
    # eager-after-find on the same shapes.  Time eager here, replay
    # after capture, and recapture once if the graph lost.
    t0 = time.perf_counter()
    for _ in range(2):
      self._eager_step_inner(static_features, static_labels,
                             global_step)
    torch.cuda.synchronize()
    t_eager = time.perf_counter() - t0

    if self._lr_schedule is not None:
      self._captured_lr = self._lr_schedule(global_step)
    grad_params = self._grad_params()
    if self.distributed:
      total = sum(p.numel() for p in grad_params)
      self._flat = torch.zeros(total, dtype=torch.float32,
                               device=self.device)
      self._assign_grad_views()

    def graph_body():
      if self.distributed:
        # grads are views of `flat`: one fill clears them all.
        self._flat.zero_()
      else:
        # set_to_none inside capture: backward then WRITES fresh
        # graph-pool buffers (stable across replays) instead of
        # zero-fill + accumulate-add per param.
        self.optimizer.zero_grad(set_to_none=True)
      with self._autocast():
        loss = self._loss_fn(static_features, static_labels)
      loss.backward()
      if not self.distributed:
        self.optimizer.step(global_step)
        if self.ema is not None:
          self.ema.update()
      return loss

    self.graphed = graph_step.GraphedTrainStep(graph_body)

    if self.distributed:
      self._check_grad_aliasing()

      def opt_body():
        self._flat.div_(float(self.world_size))
        self.optimizer.step(global_step)
        if self.ema is not None:
          self.ema.update()
        return None

      self.opt_graphed = graph_step.GraphedTrainStep(opt_body)

    # Post-capture sanity check (see comment above): replay must not be
    # slower than eager.  The decision is collective in distributed
    # mode (the replay includes no collectives, but a recapture does
    # run unsynced settle steps — every rank must take the same path).
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(2):
      self.graphed.replay()
      if self.distributed:
        dist.all_reduce(self._flat)
        self.opt_graphed.replay()
    torch.cuda.synchronize()
    t_replay = time.perf_counter() - t0
    slow = t_replay > 1.3 * t_eager + 1e-3
    if self.distributed:
      flag = torch.tensor([1.0 if slow else 0.0], device=self.device)
      dist.all_reduce(flag, op=dist.ReduceOp.MAX)
      slow = float(flag.item()) > 0.0
    if slow and attempt == 0:
      _log.warning(
          "captured graph slower than eager (%.2f vs %.2f ms/step) — "
          "find likely mid-flight at capture; recapturing",
          t_replay / 2 * 1e3, t_eager / 2 * 1e3)
      self.graphed = None
      self.opt_graphed = None
      torch.cuda.synchronize()
      self._capture(global_step, attempt=1)

  def _demote(self):
    self.graphed = None
    self.opt_graphed = None
    if self._flat is not None:
      # Eager fallback must not leave grads aliased into the comm buffer.
      for p in self.network.parameters():
        p.grad = None
      self._flat = None
    if self.distributed and self._dp_engine is None:
      from tensor2robot_amd.parallel import ddp
      self._dp_engine = ddp.DataParallelEngine(self.network)

  # -- step ------------------------------------------------------------------
  def _copy_in(self, features, labels):
    for key, t in _flatten_tensors(features).items():
      self._static_f[key].copy_(t)
    for key, t in _flatten_tensors(labels).items():
      self._static_l[key].copy_(t)

  def _eager_step_inner(self, features, labels, global_step: int):
    self.optimizer.zero_grad(set_to_none=True)
    with self._autocast():
      loss = self._loss_fn(features, labels)
    if self._dp_engine is not None:
      self._dp_engine.backward(loss)
    else:
      loss.backward()
    self.optimizer.step(global_step)
    if self.ema is not None:
      self.ema.update()
    return loss

  def _maybe_recapture(self, global_step: int):
    if self.graphed is None or self._lr_schedule is None:
      return
    lr = self._lr_schedule(global_step)
    if self._captured_lr and abs(lr - self._captured_lr) <= \
        1e-3 * abs(self._captured_lr):
      return
    self._recaptures += 1
    if self._recaptures > self._max_recaptures:
      # Continuously-decaying schedule: graphs are the wrong tool.
      _log.warning("lr schedule moves every step; demoting to eager")
      self._demote()
      return
    self._capture(global_step)
    if self.distributed:
      # The settle steps inside _capture ran without gradient sync;
      # re-align ranks.  The schedule is a deterministic function of
      # global_step, so every rank recaptures at the same step.
      for p in self.network.parameters():
        dist.broadcast(p.data, src=0)

  def step(self, features, labels, global_step: int = 0) -> torch.Tensor:
    """One training step on device-resident, preprocessed batches."""
    if not self._built:
      raise RuntimeError("call build() first")
    self._maybe_recapture(global_step)
    if self.graphed is None:
      return self._eager_step_inner(features, labels, global_step)
    self._copy_in(features, labels)
    loss = self.graphed.replay()
    if self.distributed:
      dist.all_reduce(self._flat)
      self.opt_graphed.replay()
    return loss

  @property
  def is_graphed(self) -> bool:
    return self.graphed is not None
