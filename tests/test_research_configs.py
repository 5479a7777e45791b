"""Gin-config smoke tests: every research config is a tested artifact.

Mirrors reference `utils/train_eval_test_utils.test_train_eval_gin` (§4.5):
parse the config, override step counts / sizes for CI speed, run one
train step through the full train_eval_model path.
"""

import glob
import os

import numpy as np
import pytest

from tensor2robot_amd import gin
from tensor2robot_amd.train import train_eval

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CONFIG_ROOT = os.path.join(REPO, "tensor2robot_amd", "research")

ALL_CONFIGS = sorted(
    p for p in glob.glob(os.path.join(CONFIG_ROOT, "*", "configs", "*.gin"))
    if not os.path.basename(p).startswith("common_")
    and "collect" not in os.path.basename(p))

# Per-config CI-speed overrides (applied after the file parses).
_OVERRIDES = {
    "run_train_grasping44.gin": [
        # BN in the action tower needs batch >= 2 in train mode.
        "DefaultRandomInputGenerator.batch_size = 2",
    ],
    "run_train_grasp2vec.gin": [
        "Grasp2VecModel.scene_size = (96, 96)",
        "Grasp2VecModel.goal_size = (96, 96)",
        "Grasp2VecModel.resnet_size = 18",
        "Grasp2VecPreprocessor.scene_crop = (0, 8, 96, 0, 8, 96)",
        "Grasp2VecPreprocessor.goal_crop = (0, 8, 96, 0, 8, 96)",
        "DefaultRandomInputGenerator.batch_size = 2",
    ],
    "run_train_bc_gtcond_trajectory.gin": [
        "BCZModel.image_size = (64, 64)",
        "BCZPreprocessor.image_size = (64, 64)",
    ],
    "run_train_bc_langcond_trajectory.gin": [
        "BCZModel.image_size = (64, 64)",
        "BCZPreprocessor.image_size = (64, 64)",
        "BCZPreprocessor.mock_subtask = True",
    ],
    "run_train_wtl_statespace_trial.gin": [
        "train_eval_model.input_generator_train = @DefaultRandomInputGenerator()",
        "train_eval_model.input_generator_eval = None",
        "DefaultRandomInputGenerator.batch_size = 2",
    ],
    "run_train_wtl_statespace_retrial.gin": [
        "train_eval_model.input_generator_train = @DefaultRandomInputGenerator()",
        "train_eval_model.input_generator_eval = None",
        "DefaultRandomInputGenerator.batch_size = 2",
    ],
    "run_train_wtl_vision_trial.gin": [
        "train_eval_model.input_generator_train = @DefaultRandomInputGenerator()",
        "train_eval_model.input_generator_eval = None",
        "DefaultRandomInputGenerator.batch_size = 1",
        "VRGripperEnvVisionTrialModel.episode_length = 12",
    ],
    "run_train_wtl_vision_retrial.gin": [
        "train_eval_model.input_generator_train = @DefaultRandomInputGenerator()",
        "train_eval_model.input_generator_eval = None",
        "DefaultRandomInputGenerator.batch_size = 1",
        "retrial/VRGripperEnvVisionTrialModel.episode_length = 20",
    ],
}


def test_configs_exist():
  # 15 runnable configs = full parity with the reference's config
  # surface (9 ported round 1 + the 4 WTL variants + 2 includes-backed
  # BC-Z configs; common_*.gin includes are not standalone artifacts).
  assert len(ALL_CONFIGS) >= 13


@pytest.mark.parametrize(
    "config_path", ALL_CONFIGS,
    ids=[os.path.basename(p) for p in ALL_CONFIGS])
def test_train_eval_gin(config_path, tmp_path):
  name = os.path.basename(config_path)
  overrides = ["train_eval_model.max_train_steps = 1",
               "train_eval_model.eval_steps = 1",
               f"train_eval_model.model_dir = '{tmp_path}'"]
  overrides += _OVERRIDES.get(name, [])
  gin.parse_config_files_and_bindings([config_path],
                                      "\n".join(overrides))
  result = train_eval.train_eval_model()
  assert result["global_step"] == 1
  assert np.isfinite(result["loss"])



def test_run_random_collect_gin(tmp_path):
  """The reference's collect config end-to-end: gin-parse
  run_random_collect.gin, run the actor loop, and find the TFRecords
  under root_dir/policy_collect (reference
  continuous_collect_eval_test.py:34-53)."""
  from tensor2robot_amd.utils import continuous_collect_eval
  config = os.path.join(CONFIG_ROOT, "pose_env", "configs",
                        "run_random_collect.gin")
  gin.parse_config_files_and_bindings([config], "\n".join([
      f"collect_eval_loop.root_dir = '{tmp_path}'",
      "collect_eval_loop.max_loops = 1",
      "run_meta_env.num_tasks = 2",
      "run_meta_env.num_episodes_per_adaptation = 1",
  ]))
  continuous_collect_eval.collect_eval_loop()
  records = glob.glob(os.path.join(str(tmp_path), "policy_collect",
                                   "*.tfrecord"))
  assert len(records) == 2, records
  from tensor2robot_amd.data import tfrecord
  recs = list(tfrecord.read_records(records[0]))
  assert recs and all(isinstance(r, bytes) for r in recs)
