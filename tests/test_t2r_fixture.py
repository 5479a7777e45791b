"""T2RModelFixture tests: random/recordio train, predict, golden values."""

import os

import numpy as np
import pytest
import torch

from tensor2robot_amd.research.pose_env import pose_env_models
from tensor2robot_amd.utils import mocks
from tensor2robot_amd.utils import t2r_test_fixture

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
FIXTURE = os.path.join(REPO_ROOT, "test_data", "pose_env_test_data.tfrecord")


def test_random_train(tmp_path):
  fixture = t2r_test_fixture.T2RModelFixture()
  result = fixture.random_train(mocks, "MockT2RModel",
                                model_dir=str(tmp_path))
  assert result["global_step"] == 2


def test_random_train_research_model(tmp_path):
  fixture = t2r_test_fixture.T2RModelFixture()
  result = fixture.random_train(pose_env_models, "PoseEnvRegressionModel",
                                model_dir=str(tmp_path))
  assert np.isfinite(result["loss"])


def test_recordio_train():
  fixture = t2r_test_fixture.T2RModelFixture()
  model_dir = fixture.recordio_train(pose_env_models,
                                     "PoseEnvRegressionModel", FIXTURE)
  assert os.path.exists(os.path.join(model_dir, "checkpoint"))


def test_random_predict():
  fixture = t2r_test_fixture.T2RModelFixture()
  prediction = fixture.random_predict(mocks, "MockT2RModel")
  assert prediction is not None
  assert "prediction" in prediction


def test_golden_values_roundtrip(tmp_path):
  golden_file = str(tmp_path / "golden.npy")
  fixture = t2r_test_fixture.T2RModelFixture()
  # Generate golden data, then verify against itself (determinism).
  fixture.train_and_check_golden_predictions(
      pose_env_models, "PoseEnvRegressionModel", FIXTURE, golden_file,
      generate_golden_data=True)
  assert os.path.exists(golden_file)
  fixture.train_and_check_golden_predictions(
      pose_env_models, "PoseEnvRegressionModel", FIXTURE, golden_file,
      generate_golden_data=False)


def test_train_eval_test_utils(tmp_path):
  from tensor2robot_amd.utils import train_eval_test_utils as tet
  fixture = t2r_test_fixture.T2RModelFixture()
  fixture.random_train(mocks, "MockT2RModel", model_dir=str(tmp_path))
  tet.assert_output_files(str(tmp_path))
  with pytest.raises(AssertionError):
    tet.assert_output_files(str(tmp_path), ("no_such_file_*",))


def test_train_eval_gin_helper(tmp_path):
  import os
  from tensor2robot_amd.utils import train_eval_test_utils as tet
  repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
  cfg = os.path.join(repo, "tensor2robot_amd", "research", "pose_env",
                     "configs", "run_train_reg_model.gin")
  result = tet.test_train_eval_gin(str(tmp_path), cfg)
  assert result["global_step"] == 1


def test_random_train_qtopt_e2e_model(tmp_path):
  """The reference's exact QT-Opt fixture test (t2r_models_test.py:41):
  random-train + random-predict the E2E grasping model by name."""
  from tensor2robot_amd.research.qtopt import t2r_models
  name = "Grasping44E2EOpenCloseTerminateGripperStatusHeightToBottom"
  fixture = t2r_test_fixture.T2RModelFixture()
  result = fixture.random_train(t2r_models, name,
                                model_dir=str(tmp_path))
  assert np.isfinite(result["loss"])
  prediction = fixture.random_predict(t2r_models, name)
  assert prediction is not None
