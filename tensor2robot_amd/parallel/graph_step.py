"""hipGraph-captured training step.

The QT-Opt step launches ~450 kernels (profiles/); at ~3-8 us of
dispatch overhead each, launch cost is a visible slice of a ~5 ms step.
Capturing forward+backward+optimizer+EMA into one hipGraph collapses
that to a single graph launch per step (HIP graphs are the ROCm
equivalent of CUDA graphs; torch.cuda.CUDAGraph drives hipGraph on
ROCm).

Protocol: the caller provides static input tensors and a step_fn that
reads ONLY those tensors (plus module/optimizer state).  Each iteration:
copy fresh data into the static tensors (dynamic work like on-GPU
preprocessing with host-side RNG stays OUTSIDE the graph), then
`replay()`.  Python-side state the step_fn closes over (e.g. a learning
-rate schedule) is frozen at capture time — call `recapture()` after
changing it (staircase LR decays change every ~100k steps, so this is
cheap in practice).
"""

from __future__ import annotations

import logging
from typing import Callable, Optional

import torch

_log = logging.getLogger(__name__)


class GraphedTrainStep:
  """Captures step_fn() into a hipGraph and replays it."""

  def __init__(self, step_fn: Callable[[], Optional[torch.Tensor]],
               warmup_iters: int = 3):
    self._step_fn = step_fn
    self._warmup_iters = warmup_iters
    self._graph: Optional[torch.cuda.CUDAGraph] = None
    self._static_output: Optional[torch.Tensor] = None
    self._capture()

  def _capture(self):
    import gc
    # Warm up on a side stream so allocator state and autotuned algos
    # settle before capture (the standard graph recipe).
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
      for _ in range(self._warmup_iters):
        self._step_fn()
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()

    # Quiesce the garbage collector across the capture: a GC cycle
    # mid-capture destroys dead CUDA objects (tensors, events from
    # earlier iterations) whose hipFree/hipEventDestroy on a capturing
    # stream aborts the process (observed: SIGABRT inside
    # "Garbage-collecting" during another engine's forward).
    gc.collect()
    gc.disable()
    try:
      self._graph = torch.cuda.CUDAGraph()
      with torch.cuda.graph(self._graph):
        self._static_output = self._step_fn()
    finally:
      gc.enable()
    _log.info("GraphedTrainStep: captured")

  def recapture(self):
    """Re-capture after python-side state changed (e.g. LR step)."""
    self._graph = None
    torch.cuda.synchronize()
    self._capture()

  def replay(self) -> Optional[torch.Tensor]:
    self._graph.replay()
    return self._static_output
