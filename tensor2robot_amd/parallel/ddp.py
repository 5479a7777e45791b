"""Bucketed data-parallel gradient engine over RCCL/xGMI.

The MI355X-native replacement for the reference's distributed primitives
(`tpu_model_wrapper.py:45-49,236` CrossShardOptimizer all-reduce;
`abstract_model.py:864-870` SyncReplicasOptimizer): one process per GPU,
`torch.distributed` (backend "nccl" IS RCCL on ROCm) with hand-rolled
gradient bucketing.

xGMI topology considerations: each MI355X has 7
point-to-point links at ~153 GB/s, so ring all-reduce is per-link-bound.
Fewer, larger collectives amortize per-launch cost and let RCCL's
multi-channel rings fill every link — the default bucket is therefore
128 MiB (vs torch DDP's 25 MiB), affordable with 288 GB HBM per GPU, and
buckets are laid out in reverse parameter order so reduction of early
buckets overlaps the remaining backward compute.  Optional bf16 gradient
compression halves link traffic.

CPU tests run this same engine over gloo (world_size 2).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from tensor2robot_amd import gin

_log = logging.getLogger(__name__)


class _Bucket:

  def __init__(self, params: List[torch.Tensor], comm_dtype: torch.dtype,
               device: torch.device):
    self.params = params
    self.numel = sum(p.numel() for p in params)
    self.comm_dtype = comm_dtype
    self.flat = torch.zeros(self.numel, dtype=comm_dtype, device=device)
    self.offsets: Dict[int, int] = {}
    offset = 0
    for p in params:
      self.offsets[id(p)] = offset
      offset += p.numel()
    self.pending = 0
    self.work: Optional[dist.Work] = None

  def reset(self):
    self.pending = len(self.params)
    self.work = None


@gin.configurable
class DataParallelEngine:
  """Registers grad hooks on a module; overlaps all-reduce with backward."""

  def __init__(self, module: torch.nn.Module,
               bucket_cap_mb: float = 128.0,
               gradient_compression: str = "none",
               process_group=None):
    if not dist.is_available() or not dist.is_initialized():
      raise RuntimeError(
          "torch.distributed must be initialized before DataParallelEngine")
    self.module = module
    self.group = process_group
    self.world_size = dist.get_world_size(self.group)
    if gradient_compression not in ("none", "bf16"):
      raise ValueError(f"Unknown compression {gradient_compression}")
    self._compression = gradient_compression
    self._bucket_cap = int(bucket_cap_mb * 1024 * 1024)
    self._params = [p for p in module.parameters() if p.requires_grad]
    self._sync_initial_state()
    self._buckets = self._build_buckets()
    self._bucket_of: Dict[int, _Bucket] = {}
    for bucket in self._buckets:
      for p in bucket.params:
        self._bucket_of[id(p)] = bucket
    self._hooks = []
    for p in self._params:
      self._hooks.append(
          p.register_post_accumulate_grad_hook(self._on_grad_ready))
    self._active = False

  # -- construction --------------------------------------------------------
  def _sync_initial_state(self):
    """Broadcast rank-0 weights so every replica starts identical."""
    with torch.no_grad():
      for t in self.module.state_dict().values():
        if isinstance(t, torch.Tensor) and t.numel() > 0 and \
            t.dtype.is_floating_point:
          dist.broadcast(t, src=0, group=self.group)

  def _comm_dtype(self, param_dtype: torch.dtype) -> torch.dtype:
    if self._compression == "bf16" and param_dtype == torch.float32:
      return torch.bfloat16
    return param_dtype

  def _build_buckets(self) -> List[_Bucket]:
    """Reverse registration order ~= backward completion order."""
    buckets: List[_Bucket] = []
    current: List[torch.Tensor] = []
    current_bytes = 0
    for p in reversed(self._params):
      current.append(p)
      current_bytes += p.numel() * p.element_size()
      if current_bytes >= self._bucket_cap:
        buckets.append(self._make_bucket(current))
        current, current_bytes = [], 0
    if current:
      buckets.append(self._make_bucket(current))
    _log.info("DataParallelEngine: %d params in %d buckets (cap %.0f MiB)",
              len(self._params), len(buckets),
              self._bucket_cap / (1024 * 1024))
    return buckets

  def _make_bucket(self, params: List[torch.Tensor]) -> _Bucket:
    dtype = self._comm_dtype(params[0].dtype)
    return _Bucket(list(params), dtype, params[0].device)

  # -- backward ------------------------------------------------------------
  def backward(self, loss: torch.Tensor):
    """loss.backward() with gradient all-reduce overlapped + finalized."""
    for bucket in self._buckets:
      bucket.reset()
    self._active = True
    try:
      loss.backward()
    finally:
      self._finalize()
      self._active = False

  def _on_grad_ready(self, param: torch.Tensor):
    if not self._active:
      return
    bucket = self._bucket_of[id(param)]
    offset = bucket.offsets[id(param)]
    grad = param.grad
    bucket.flat[offset: offset + grad.numel()].copy_(
        grad.detach().reshape(-1).to(bucket.comm_dtype))
    bucket.pending -= 1
    if bucket.pending == 0:
      self._launch(bucket)

  def _launch(self, bucket: _Bucket):
    bucket.flat.div_(self.world_size)
    bucket.work = dist.all_reduce(bucket.flat, op=dist.ReduceOp.SUM,
                                  group=self.group, async_op=True)

  def _finalize(self):
    for bucket in self._buckets:
      if bucket.work is None and bucket.pending > 0:
        # Some params never produced grads this step: zero their slots,
        # reduce what we have (keeps ranks collective-aligned).
        for p in bucket.params:
          if p.grad is None:
            offset = bucket.offsets[id(p)]
            bucket.flat[offset: offset + p.numel()].zero_()
        self._launch(bucket)
    for bucket in self._buckets:
      if bucket.work is not None:
        bucket.work.wait()
      for p in bucket.params:
        offset = bucket.offsets[id(p)]
        reduced = bucket.flat[offset: offset + p.numel()].reshape(
            p.shape).to(p.dtype)
        if p.grad is None:
          p.grad = reduced.clone()
        else:
          p.grad.copy_(reduced)

  def detach(self):
    for h in self._hooks:
      h.remove()
    self._hooks = []


def all_reduce_scalar(value: float, average: bool = True) -> float:
  """Scalar metric reduction across the DP group.  The tensor lives on
  the GPU when the backend is nccl/RCCL (which rejects CPU tensors)."""
  if not dist.is_available() or not dist.is_initialized():
    return value
  device = "cuda" if dist.get_backend() == "nccl" and       torch.cuda.is_available() else "cpu"
  t = torch.tensor([value], dtype=torch.float64, device=device)
  dist.all_reduce(t)
  if average:
    t /= dist.get_world_size()
  return float(t.item())
