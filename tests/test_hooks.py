"""Hooks suite: async export handshake, TD3 lagged exports, GC, loggers.

Mirrors reference async_export_hook_builder_tpu_test / td3_test /
checkpoint_hooks_test (§4.8): the actor/learner contract is the
filesystem, tested end-to-end in one process.
"""

import os
import time

import numpy as np
import pytest
import torch

from tensor2robot_amd.models import optimizers
from tensor2robot_amd.predictors import exported_savedmodel_predictor as esp
from tensor2robot_amd.train import hooks as hooks_mod
from tensor2robot_amd.train import train_eval
from tensor2robot_amd.utils import mocks


def _make_model():
  return mocks.MockT2RModel(
      device_type="cpu",
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-2))


def test_directory_version_gc(tmp_path):
  for v in [100, 200, 300, 400, 500]:
    os.makedirs(tmp_path / str(v))
  (tmp_path / "temp-600").mkdir()  # non-numeric survives
  gc = hooks_mod._DirectoryVersionGC(str(tmp_path), keep=2)
  gc.collect()
  left = sorted(os.listdir(tmp_path))
  assert left == ["400", "500", "temp-600"]


def test_async_export_then_predictor_restore(tmp_path):
  """Trainer exports on a timer; a polling predictor picks it up."""
  model = _make_model()
  gen = mocks.MockInputGenerator(batch_size=8)
  builder = hooks_mod.AsyncExportHookBuilder(
      export_dir=str(tmp_path / "export"), save_secs=0.0)  # every step
  result = train_eval.train_eval_model(
      t2r_model=model, input_generator_train=gen,
      input_generator_eval=None, max_train_steps=3,
      model_dir=str(tmp_path), train_hook_builders=[builder])
  assert result["global_step"] == 3
  versions = [d for d in os.listdir(tmp_path / "export") if d.isdigit()]
  assert versions, "async export produced nothing"
  predictor = esp.ExportedSavedModelPredictor(str(tmp_path / "export"),
                                              timeout=5)
  assert predictor.restore()
  out = predictor.predict(
      {"measured_position": np.ones((2, 3), np.float32)})
  assert out["prediction"].shape == (2, 1)


def test_td3_lagged_export(tmp_path):
  """The lagged dir always holds the PREVIOUS export (TD3 targets)."""
  model = _make_model()
  gen = mocks.MockInputGenerator(batch_size=8)
  builder = hooks_mod.TD3Hooks(export_dir=str(tmp_path / "export"),
                               lagged_export_dir=str(tmp_path / "lag"),
                               save_secs=0.0)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=gen,
      input_generator_eval=None, max_train_steps=4,
      model_dir=str(tmp_path), train_hook_builders=[builder])
  exports = sorted(int(d) for d in os.listdir(tmp_path / "export")
                   if d.isdigit())
  assert len(exports) >= 2
  lagged = sorted(int(d) for d in os.listdir(tmp_path / "lag")
                  if d.isdigit())
  assert lagged, "no lagged export"
  # Every lagged version is an OLDER export version.
  assert max(lagged) < max(exports)
  # Lagged copies are loadable servables too.
  predictor = esp.ExportedSavedModelPredictor(str(tmp_path / "lag"),
                                              timeout=5)
  assert predictor.restore()


def test_lagged_hook_resync_on_restart(tmp_path):
  export_dir = tmp_path / "export"
  for v in ["100", "200"]:
    (export_dir / v).mkdir(parents=True)
  hook = hooks_mod.LaggedCheckpointExportHook(
      export_fn=lambda context, path: None,
      export_dir=str(export_dir), lagged_export_dir=str(tmp_path / "lag"))
  # Resync picked the newest existing export as "previous".
  hook.after_checkpoint(context=None, checkpoint_path="unused")
  assert sorted(os.listdir(tmp_path / "lag")) == ["200"]


def test_gin_config_logger_and_variable_logger(tmp_path):
  model = _make_model()
  gen = mocks.MockInputGenerator(batch_size=8)

  class _Builder(hooks_mod.HookBuilder):

    def create_hooks(self, t2r_model, trainer):
      return [hooks_mod.GinConfigLoggerHook(),
              hooks_mod.VariableLoggerHook(max_num_variable_values=2)]

  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=gen,
      input_generator_eval=None, max_train_steps=2,
      model_dir=str(tmp_path), train_hook_builders=[_Builder()])
  assert os.path.exists(tmp_path / "operative_config-0.gin")
