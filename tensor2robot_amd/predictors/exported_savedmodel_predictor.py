"""Predictor over exported servable directories (poll + load + predict).

Reference: `predictors/exported_savedmodel_predictor.py:53-360` — polls the
export root for the newest numeric subdir (skipping temp- dirs), busy-waits
up to `timeout`, loads the t2r_assets specs + servable, optionally restores
on a background thread.
"""

from __future__ import annotations

import enum
import json
import logging
import os
import threading
import time
from typing import Dict, List, Optional

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.export_generators import abstract_export_generator as aeg
from tensor2robot_amd.predictors import abstract_predictor
from tensor2robot_amd.specs import tensorspec_utils as tsu

_log = logging.getLogger(__name__)

_BUSY_WAIT_SLEEP = 1.0


@gin.constants_from_enum
class RestoreOptions(enum.Enum):
  DO_NOT_RESTORE = 0
  RESTORE_SYNCHRONOUSLY = 1
  RESTORE_ASYNCHRONOUSLY = 2


def latest_valid_model_dirs(export_dir: str) -> List[str]:
  """Numeric subdirs sorted oldest->newest, skipping temp dirs (ref :314)."""
  if not os.path.isdir(export_dir):
    return []
  dirs = []
  for name in os.listdir(export_dir):
    path = os.path.join(export_dir, name)
    if not name.isdigit() or not os.path.isdir(path):
      continue
    if not os.path.exists(os.path.join(path, aeg.SERVABLE_NAME)):
      continue  # still being written
    dirs.append((int(name), path))
  return [p for _, p in sorted(dirs)]


@gin.configurable
class ExportedSavedModelPredictor(abstract_predictor.AbstractPredictor):

  def __init__(self, export_dir: str, timeout: float = 600.0,
               device: str = "cpu",
               restore_model_option: RestoreOptions =
               RestoreOptions.RESTORE_SYNCHRONOUSLY):
    self._export_dir = export_dir
    self._timeout = timeout
    self._device = torch.device(device)
    self._module = None
    self._metadata = None
    self._assets: Optional[tsu.T2RAssets] = None
    self._model_path: Optional[str] = None
    self._parse_fn = None
    self._restore_thread: Optional[threading.Thread] = None
    self._lock = threading.Lock()
    if restore_model_option == RestoreOptions.RESTORE_SYNCHRONOUSLY:
      self.restore()
    elif restore_model_option == RestoreOptions.RESTORE_ASYNCHRONOUSLY:
      self.restore_async()

  def restore_async(self):
    self._restore_thread = threading.Thread(target=self.restore,
                                            daemon=True)
    self._restore_thread.start()

  def _wait_for_restore(self):
    if self._restore_thread is not None and \
        self._restore_thread.is_alive():
      self._restore_thread.join()

  def restore(self) -> bool:
    """Loads the newest export, busy-waiting up to timeout (ref :181-272)."""
    deadline = time.time() + self._timeout
    while True:
      dirs = latest_valid_model_dirs(self._export_dir)
      if dirs:
        newest = dirs[-1]
        if newest != self._model_path:
          if self._load(newest):
            return True
        else:
          return True  # already current
      if time.time() >= deadline:
        _log.warning("No loadable export in %s after %.0fs",
                     self._export_dir, self._timeout)
        return False
      time.sleep(_BUSY_WAIT_SLEEP)

  def _load(self, model_dir: str) -> bool:
    try:
      with open(os.path.join(model_dir, aeg.METADATA_NAME)) as f:
        metadata = json.load(f)
      assets_path = os.path.join(model_dir, tsu.EXTRA_ASSETS_DIRECTORY,
                                 tsu.T2R_ASSETS_FILENAME)
      assets = tsu.load_t2r_assets_from_file(assets_path)
      servable_path = os.path.join(model_dir, aeg.SERVABLE_NAME)
      if metadata.get("format") == "torchscript":
        module = torch.jit.load(servable_path, map_location=self._device)
      else:
        module = torch.load(servable_path, map_location=self._device,
                            weights_only=False)
      module.eval()
      with self._lock:
        self._module = module
        self._metadata = metadata
        self._assets = assets
        self._model_path = model_dir
      return True
    except (OSError, ValueError, RuntimeError) as e:
      _log.warning("Failed to load export %s: %s", model_dir, e)
      return False

  @property
  def is_loaded(self) -> bool:
    self._wait_for_restore()
    return self._module is not None

  @property
  def global_step(self) -> int:
    self._wait_for_restore()
    return self._assets.global_step if self._assets else -1

  @property
  def model_path(self) -> Optional[str]:
    return self._model_path

  def get_feature_specification(self):
    self._wait_for_restore()
    return self._assets.feature_spec if self._assets else None

  def get_label_specification(self):
    self._wait_for_restore()
    return self._assets.label_spec if self._assets else None

  def predict(self, features: Dict[str, np.ndarray]
              ) -> Dict[str, np.ndarray]:
    self.assert_is_loaded()
    spec = self.get_feature_specification()
    in_keys = self._metadata["in_keys"]
    tensors = []
    for key in in_keys:
      value = features[key]
      t = torch.as_tensor(np.asarray(value))
      key_spec = spec.get(key) if spec is not None else None
      if key_spec is not None:
        # Action-tile auto-expand (reference :106-117): if the spec has an
        # extra leading sample dim relative to the feed, expand.
        want_rank = len(key_spec.shape) + 1  # + batch
        if t.dim() == want_rank - 1:
          t = t.unsqueeze(0)
        if key_spec.dtype.is_floating_point and not \
            t.dtype.is_floating_point:
          t = t.float()
        elif key_spec.dtype.is_floating_point:
          t = t.to(torch.float32)
      tensors.append(t.to(self._device))
    with torch.no_grad():
      outputs = self._module(*tensors)
    if isinstance(outputs, torch.Tensor):
      outputs = (outputs,)
    out_keys = self._metadata["out_keys"]
    return {k: v.detach().float().cpu().numpy()
            for k, v in zip(out_keys, outputs)}

  def predict_serialized(self, serialized_examples) -> Dict[str, np.ndarray]:
    """Serving from serialized tf.Example bytes (reference tf_example
    receiver, default_export_generator.py:84-133): the assets' feature
    spec drives the auto-parser, then the numpy path runs."""
    self.assert_is_loaded()
    from tensor2robot_amd.data import parser as parser_mod
    spec = self.get_feature_specification()
    parse_fn = parser_mod.create_parse_example_fn(spec)
    if not isinstance(serialized_examples, dict):
      serialized_examples = {"": list(serialized_examples)}
    features, _ = parse_fn(serialized_examples)
    return self.predict({k: v.numpy() if hasattr(v, "numpy") else v
                         for k, v in features.items()})

  def close(self):
    self._wait_for_restore()
    self._module = None
