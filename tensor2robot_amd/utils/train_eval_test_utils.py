"""Assertions shared by train/eval integration tests.

Reference `utils/train_eval_test_utils.py`: DEFAULT_TRAIN_FILENAME_PATTERNS
:26 (checkpoint files, operative gin, graph), assert_output_files :33,
test_train_eval_gin :68 (parameterized gin-config smoke runner).
"""

from __future__ import annotations

import glob
import os
from typing import Iterable, Optional

from tensor2robot_amd import gin
from tensor2robot_amd.train import train_eval

DEFAULT_TRAIN_FILENAME_PATTERNS = (
    "checkpoint",
    "model.ckpt-*.pt",
    "operative_config-*.gin",
    "events.jsonl",
)


def assert_output_files(model_dir: str,
                        expected_output_filename_patterns: Iterable[str] =
                        DEFAULT_TRAIN_FILENAME_PATTERNS):
  """Asserts the training artifacts exist (reference :33-64)."""
  missing = []
  for pattern in expected_output_filename_patterns:
    if not glob.glob(os.path.join(model_dir, pattern)):
      missing.append(pattern)
  assert not missing, (
      f"Missing train outputs {missing} in {model_dir}: "
      f"{sorted(os.listdir(model_dir))}")


def test_train_eval_gin(model_dir: str, full_gin_path: str,
                        max_train_steps: int = 1, eval_steps: int = 1,
                        extra_bindings: Optional[str] = None):
  """Parses and runs a gin config for a few steps (reference :68-...)."""
  bindings = [f"train_eval_model.max_train_steps = {max_train_steps}",
              f"train_eval_model.eval_steps = {eval_steps}",
              f"train_eval_model.model_dir = '{model_dir}'"]
  if extra_bindings:
    bindings.append(extra_bindings)
  gin.parse_config_files_and_bindings([full_gin_path],
                                      "\n".join(bindings))
  result = train_eval.train_eval_model()
  assert result["global_step"] == max_train_steps
  assert_output_files(model_dir,
                      ("checkpoint", "model.ckpt-*.pt", "events.jsonl"))
  return result
