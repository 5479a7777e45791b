"""Ensemble predictor over multiple export roots (reference ensemble_...:32).

N ExportedSavedModelPredictors over comma-separated export dirs; per
`reset()` a random subset of `ensemble_size` is selected; `predict` averages
their outputs.  `local_export_root` remaps paths for on-robot deploys
(reference :49-57).
"""

from __future__ import annotations

import os
import random
from typing import Dict, List, Optional

import numpy as np

from tensor2robot_amd import gin
from tensor2robot_amd.predictors import abstract_predictor
from tensor2robot_amd.predictors import exported_savedmodel_predictor as esp


@gin.configurable
class EnsembleExportedSavedModelPredictor(
    abstract_predictor.AbstractPredictor):

  def __init__(self, export_dirs: str, ensemble_size: int = 1,
               local_export_root: Optional[str] = None,
               timeout: float = 600.0, seed: Optional[int] = None):
    dirs = [d.strip() for d in export_dirs.split(",") if d.strip()]
    if local_export_root:
      dirs = [os.path.join(local_export_root, os.path.basename(d))
              for d in dirs]
    self._predictors = [
        esp.ExportedSavedModelPredictor(
            d, timeout=timeout,
            restore_model_option=esp.RestoreOptions.DO_NOT_RESTORE)
        for d in dirs
    ]
    self._ensemble_size = min(ensemble_size, len(self._predictors))
    self._rng = random.Random(seed)
    self._active: List = []
    self.reset()

  def reset(self):
    self._active = self._rng.sample(self._predictors, self._ensemble_size)

  def restore(self) -> bool:
    ok = True
    for p in self._predictors:
      ok = p.restore() and ok
    return ok

  @property
  def is_loaded(self) -> bool:
    return all(p.is_loaded for p in self._active)

  @property
  def global_step(self) -> int:
    steps = [p.global_step for p in self._active if p.is_loaded]
    return min(steps) if steps else -1

  @property
  def model_path(self):
    return ",".join(p.model_path or "" for p in self._active)

  def get_feature_specification(self):
    return self._active[0].get_feature_specification()

  def predict(self, features: Dict[str, np.ndarray]
              ) -> Dict[str, np.ndarray]:
    outputs = [p.predict(features) for p in self._active]
    out: Dict[str, np.ndarray] = {}
    for key in outputs[0]:
      out[key] = np.mean([o[key] for o in outputs], axis=0)
    return out

  def close(self):
    for p in self._predictors:
      p.close()
