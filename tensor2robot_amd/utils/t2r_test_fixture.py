"""T2RModelFixture: 2-step hermetic model tests on spec-random data.

Reference `utils/t2r_test_fixture.py:42-196`: random_train :57 (2 steps,
batch 2, spec-random inputs), recordio_train :87, random_predict :115,
train_and_check_golden_predictions :143 (fixed seed -> golden_values.npy
numpy comparison at decimal=5 — the numerical-regression harness that
guards data -> checkpoint determinism across refactors).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.data import input_generators
from tensor2robot_amd.train import hooks as hooks_mod
from tensor2robot_amd.train import train_eval

MAX_TRAIN_STEPS = 2
BATCH_SIZE = 2


class T2RModelFixture:
  """Quick-train harness for any T2RModel (reference :42)."""

  def __init__(self, test_case=None, extra_bindings=None):
    self._test_case = test_case
    if extra_bindings:
      for parameter, binding in extra_bindings.items():
        gin.bind_parameter(parameter, binding)

  def _tmpdir(self, model_dir: Optional[str]) -> str:
    if model_dir:
      return model_dir
    if self._test_case is not None and hasattr(self._test_case,
                                               "create_tempdir"):
      return self._test_case.create_tempdir().full_path
    import tempfile
    return tempfile.mkdtemp(prefix="t2r_fixture_")

  def random_train(self, module, model_name: str, model_dir=None,
                   batch_size: int = BATCH_SIZE,
                   max_train_steps: int = MAX_TRAIN_STEPS,
                   **module_kwargs):
    """Instantiates and trains a model on spec-random inputs (ref :57)."""
    t2r_model = getattr(module, model_name)(**module_kwargs)
    return self.random_train_model(t2r_model, model_dir=model_dir,
                                   batch_size=batch_size,
                                   max_train_steps=max_train_steps)

  def random_train_model(self, t2r_model, model_dir=None,
                         batch_size: int = BATCH_SIZE,
                         max_train_steps: int = MAX_TRAIN_STEPS):
    model_dir = self._tmpdir(model_dir)
    gen = input_generators.DefaultRandomInputGenerator(
        batch_size=batch_size)
    result = train_eval.train_eval_model(
        t2r_model=t2r_model, input_generator_train=gen,
        input_generator_eval=None, max_train_steps=max_train_steps,
        model_dir=model_dir)
    assert result["global_step"] == max_train_steps
    assert os.path.exists(os.path.join(model_dir, "checkpoint"))
    return result

  def recordio_train(self, module, model_name: str, file_patterns: str,
                     model_dir=None, batch_size: int = BATCH_SIZE,
                     max_train_steps: int = MAX_TRAIN_STEPS,
                     train_hook_builders=None, **module_kwargs):
    """Trains on TFRecord data for a few steps (reference :87)."""
    t2r_model = getattr(module, model_name)(**module_kwargs)
    model_dir = self._tmpdir(model_dir)
    gen = input_generators.DefaultRecordInputGenerator(
        file_patterns=file_patterns, batch_size=batch_size, seed=123)
    result = train_eval.train_eval_model(
        t2r_model=t2r_model, input_generator_train=gen,
        input_generator_eval=None, max_train_steps=max_train_steps,
        model_dir=model_dir, train_hook_builders=train_hook_builders)
    assert result["global_step"] == max_train_steps
    return model_dir

  def random_predict(self, module, model_name: str, batch_size: int = 1,
                     **module_kwargs):
    """One prediction on spec-random inputs (reference :115)."""
    t2r_model = getattr(module, model_name)(**module_kwargs)
    gen = input_generators.DefaultRandomInputGenerator(
        batch_size=batch_size, max_batches=1)
    for prediction in train_eval.predict_from_model(
        t2r_model=t2r_model, input_generator=gen,
        model_dir=self._tmpdir(None)):
      return prediction
    return None

  def train_and_check_golden_predictions(
      self, module, model_name: str, file_patterns: str,
      golden_data_filename: str, generate_golden_data: bool = False,
      decimal: int = 5, **module_kwargs):
    """Golden-value numerical regression (reference :143-196)."""
    torch.manual_seed(123)
    np.random.seed(123)
    model_dir = self.recordio_train(
        module, model_name, file_patterns,
        train_hook_builders=[hooks_mod.GoldenValuesHookBuilder()],
        **module_kwargs)
    golden_path = os.path.join(model_dir, "golden_values.npy")
    model_data = np.load(golden_path, allow_pickle=True)
    if generate_golden_data:
      with open(golden_data_filename, "wb") as f:
        np.save(f, model_data)
      return model_data
    golden_data = np.load(golden_data_filename, allow_pickle=True)
    golden = golden_data.item() if golden_data.shape == () else \
        golden_data[0]
    actual = model_data.item() if model_data.shape == () else model_data[0]
    for key, golden_value in golden.items():
      if key not in actual:
        continue
      np.testing.assert_almost_equal(actual[key], golden_value,
                                     decimal=decimal)
    return model_data
