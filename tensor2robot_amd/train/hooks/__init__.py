"""Hook protocol + builders for the native train loop.

Reference: `hooks/hook_builder.py:27-44` (HookBuilder.create_hooks),
`hooks/checkpoint_hooks.py` (export-after-save + lagged TD3 export + dir GC),
`hooks/async_export_hook_builder.py` (periodic async export),
`hooks/golden_values_hook_builder.py` (golden-value regression capture),
`hooks/gin_config_hook_builder.py`, `hooks/variable_logger_hook.py`.

A TrainHook sees the loop through a small context object; hooks run on the
chief rank only unless `every_rank` is True.
"""

from __future__ import annotations

import abc
import logging
import os
import shutil
import threading
import time
from typing import Callable, Dict, List, Optional

import numpy as np
import torch

from tensor2robot_amd import gin

_log = logging.getLogger(__name__)


class TrainHook(abc.ABC):
  """Side-band service on the train loop (SessionRunHook analog)."""

  every_rank = False

  def begin(self, context):
    pass

  def before_step(self, context):
    pass

  def after_step(self, context, outputs):
    pass

  def after_checkpoint(self, context, checkpoint_path: str):
    pass

  def end(self, context):
    pass


@gin.configurable
class HookBuilder:
  """Creates hooks for a (model, trainer) pair (reference :27-44)."""

  def create_hooks(self, t2r_model, trainer) -> List[TrainHook]:
    del t2r_model, trainer
    return []


class _DirectoryVersionGC:
  """Keeps the newest N numeric subdirectories (reference :31-48)."""

  def __init__(self, root: str, keep: int = 3):
    self._root = root
    self._keep = keep

  def collect(self):
    if not os.path.isdir(self._root):
      return
    versions = []
    for name in os.listdir(self._root):
      if name.isdigit() and os.path.isdir(os.path.join(self._root, name)):
        versions.append(int(name))
    for version in sorted(versions)[: -self._keep] if len(versions) > \
        self._keep else []:
      shutil.rmtree(os.path.join(self._root, str(version)),
                    ignore_errors=True)


class CheckpointExportHook(TrainHook):
  """Exports a servable after every checkpoint save (reference :51-88)."""

  def __init__(self, export_fn: Callable[[object, str], str],
               export_dir: str, keep_versions: int = 3):
    self._export_fn = export_fn
    self._export_dir = export_dir
    self._gc = _DirectoryVersionGC(export_dir, keep_versions)

  def after_checkpoint(self, context, checkpoint_path):
    self._export_fn(context, checkpoint_path)
    self._gc.collect()


class LaggedCheckpointExportHook(CheckpointExportHook):
  """Also maintains a one-version-lagged export copy (TD3 targets).

  Reference `hooks/checkpoint_hooks.py:91-201`: `lagged_export_dir` always
  holds the PREVIOUS export so TD3-style consumers can read a delayed
  target network; resyncs on startup.
  """

  def __init__(self, export_fn, export_dir: str, lagged_export_dir: str,
               keep_versions: int = 3):
    super().__init__(export_fn, export_dir, keep_versions)
    self._lagged_dir = lagged_export_dir
    self._lag_gc = _DirectoryVersionGC(lagged_export_dir, keep_versions)
    self._previous_export: Optional[str] = None
    self._resync()

  def _resync(self):
    """On restart: lagged dir should hold the newest existing export."""
    if not os.path.isdir(self._export_dir):
      return
    versions = sorted(int(n) for n in os.listdir(self._export_dir)
                      if n.isdigit())
    if versions:
      self._previous_export = os.path.join(self._export_dir,
                                           str(versions[-1]))

  def after_checkpoint(self, context, checkpoint_path):
    if self._previous_export is not None and \
        os.path.isdir(self._previous_export):
      os.makedirs(self._lagged_dir, exist_ok=True)
      dst = os.path.join(self._lagged_dir,
                         os.path.basename(self._previous_export))
      if not os.path.exists(dst):
        tmp = dst + "_temp"
        shutil.copytree(self._previous_export, tmp)
        os.replace(tmp, dst)
      self._lag_gc.collect()
    before = set(os.listdir(self._export_dir)) \
        if os.path.isdir(self._export_dir) else set()
    self._export_fn(context, checkpoint_path)
    after = set(os.listdir(self._export_dir)) \
        if os.path.isdir(self._export_dir) else set()
    new = sorted(after - before)
    if new:
      self._previous_export = os.path.join(self._export_dir, new[-1])
    self._gc.collect()


class AsyncCheckpointHook(TrainHook):
  """Periodic (save_secs) checkpoint + export, off the critical path.

  Reference `hooks/async_export_hook_builder.py:87-134`
  (AsyncCheckpointSaverHook(save_secs=90) + export listener).  The save
  itself snapshots weights synchronously (cheap) and runs the export
  callback on a worker thread.
  """

  def __init__(self, save_secs: float = 90.0,
               export_fn: Optional[Callable] = None,
               keep_versions: int = 3, export_dir: Optional[str] = None):
    self._save_secs = save_secs
    self._export_fn = export_fn
    self._last_save = 0.0
    self._worker: Optional[threading.Thread] = None
    self._gc = _DirectoryVersionGC(export_dir, keep_versions) \
        if export_dir else None

  def begin(self, context):
    self._last_save = time.time()

  def after_step(self, context, outputs):
    now = time.time()
    if now - self._last_save < self._save_secs:
      return
    self._last_save = now
    path = context.save_checkpoint()
    if self._export_fn is not None:
      if self._worker is not None and self._worker.is_alive():
        self._worker.join()
      self._worker = threading.Thread(
          target=self._run_export, args=(context, path), daemon=True)
      self._worker.start()

  def _run_export(self, context, path):
    try:
      self._export_fn(context, path)
      if self._gc is not None:
        self._gc.collect()
    except Exception:  # pragma: no cover
      _log.exception("Async export failed")

  def end(self, context):
    if self._worker is not None and self._worker.is_alive():
      self._worker.join(timeout=60)


GOLDEN_COLLECTION: Dict[str, torch.Tensor] = {}


def add_golden_tensor(name: str, tensor: torch.Tensor):
  """Models register tensors for golden-value regression (reference :37-39)."""
  GOLDEN_COLLECTION[name] = tensor.detach().float().cpu()


class GoldenValuesHook(TrainHook):
  """Fetches registered golden tensors each step, saves golden_values.npy.

  Reference `hooks/golden_values_hook_builder.py:37-79` — protects
  data->checkpoint determinism across refactors.
  """

  def __init__(self, log_dir: str):
    self._log_dir = log_dir
    self._values: List[Dict[str, np.ndarray]] = []

  def after_step(self, context, outputs):
    step_values = {k: v.numpy().copy()
                   for k, v in GOLDEN_COLLECTION.items()}
    GOLDEN_COLLECTION.clear()
    # The loss is always a golden value: even models that register no
    # tensors get data->checkpoint determinism coverage.
    loss = getattr(outputs, "loss", None)
    if loss is None and isinstance(outputs, dict):
      loss = outputs.get("loss")
    if loss is not None:
      step_values.setdefault(
          "loss", np.asarray(float(loss), dtype=np.float64))
    if step_values:
      self._values.append(step_values)

  def end(self, context):
    os.makedirs(self._log_dir, exist_ok=True)
    path = os.path.join(self._log_dir, "golden_values.npy")
    np.save(path, np.asarray(self._values, dtype=object),
            allow_pickle=True)


@gin.configurable
class GoldenValuesHookBuilder(HookBuilder):

  def __init__(self, log_dir: Optional[str] = None):
    self._log_dir = log_dir

  def create_hooks(self, t2r_model, trainer):
    return [GoldenValuesHook(self._log_dir or trainer.model_dir)]


class GinConfigLoggerHook(TrainHook):
  """Logs + saves the operative gin config (reference gin_config_hook)."""

  def __init__(self, log_dir: Optional[str] = None):
    self._log_dir = log_dir
    self._done = False

  def begin(self, context):
    if self._done:
      return
    self._done = True
    config = gin.operative_config_str()
    _log.info("Operative gin config:\n%s", config)
    log_dir = self._log_dir or context.model_dir
    if log_dir:
      os.makedirs(log_dir, exist_ok=True)
      with open(os.path.join(log_dir, "operative_config-0.gin"), "w") as f:
        f.write(config)


class VariableLoggerHook(TrainHook):
  """Logs mean/std of all parameters per step (debug; reference :27-62)."""

  def __init__(self, max_num_variable_values: Optional[int] = None):
    self._max_values = max_num_variable_values

  def after_step(self, context, outputs):
    for name, p in context.network.named_parameters():
      data = p.detach().float()
      msg = (f"step={context.global_step} var={name} "
             f"mean={data.mean().item():.6f} std={data.std().item():.6f}")
      if self._max_values:
        flat = data.flatten()[: self._max_values]
        msg += f" values={flat.cpu().numpy()}"
      _log.info(msg)


@gin.configurable
class AsyncExportHookBuilder(HookBuilder):
  """Builds the periodic async export hook (reference :87-134)."""

  def __init__(self, export_dir: Optional[str] = None,
               save_secs: float = 90.0, keep_versions: int = 3,
               create_export_fn=None):
    self._export_dir = export_dir
    self._save_secs = save_secs
    self._keep_versions = keep_versions
    self._create_export_fn = create_export_fn

  def create_hooks(self, t2r_model, trainer):
    export_dir = self._export_dir or os.path.join(trainer.model_dir,
                                                  "export")
    export_fn = self._create_export_fn or default_create_export_fn(
        export_dir)
    return [AsyncCheckpointHook(save_secs=self._save_secs,
                                export_fn=export_fn,
                                keep_versions=self._keep_versions,
                                export_dir=export_dir)]


def default_create_export_fn(export_dir: str, export_generator=None):
  """Standard export callback: servable + t2r_assets (reference :42-83)."""

  def export_fn(context, checkpoint_path: str) -> str:
    from tensor2robot_amd.export_generators import default_export_generator
    gen = export_generator or \
        default_export_generator.DefaultExportGenerator()
    gen.set_specification_from_model(context.model)
    return gen.export(context.model, export_dir,
                      global_step=context.global_step)

  return export_fn


@gin.configurable
class TD3Hooks(HookBuilder):
  """Async export + lagged export dir (TD3 target networks as servables).

  Reference `hooks/td3.py:37-132`.
  """

  def __init__(self, export_dir: Optional[str] = None,
               lagged_export_dir: Optional[str] = None,
               save_secs: float = 90.0, keep_versions: int = 3):
    self._export_dir = export_dir
    self._lagged_export_dir = lagged_export_dir
    self._save_secs = save_secs
    self._keep_versions = keep_versions

  def create_hooks(self, t2r_model, trainer):
    export_dir = self._export_dir or os.path.join(trainer.model_dir,
                                                  "export")
    lagged_dir = self._lagged_export_dir or os.path.join(
        trainer.model_dir, "lagged_export")
    export_fn = default_create_export_fn(export_dir)
    lagged = LaggedCheckpointExportHook(
        export_fn, export_dir, lagged_dir,
        keep_versions=self._keep_versions)
    periodic = AsyncCheckpointHook(save_secs=self._save_secs,
                                   export_fn=None)

    # Chain: periodic checkpoint triggers lagged export via after_checkpoint.
    class _Chain(TrainHook):

      def begin(self, context):
        periodic.begin(context)

      def after_step(self, context, outputs):
        now = time.time()
        if now - periodic._last_save < periodic._save_secs:
          return
        periodic._last_save = now
        path = context.save_checkpoint()
        lagged.after_checkpoint(context, path)

      def end(self, context):
        periodic.end(context)

    return [_Chain()]


# Reference class names (hooks/checkpoint_hooks.py:51,91 and
# hooks/gin_config_hook_builder.py): the torch-native implementations
# above under the names a reference user would import.
CheckpointExportListener = CheckpointExportHook
LaggedCheckpointListener = LaggedCheckpointExportHook


class OperativeGinConfigLoggerHookBuilder(HookBuilder):
  """Builds GinConfigLoggerHook (reference gin_config_hook_builder.py)."""

  def create_hooks(self, t2r_model, trainer):
    return [GinConfigLoggerHook()]
