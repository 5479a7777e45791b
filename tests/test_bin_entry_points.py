"""The CLI binaries end to end (reference bin/run_t2r_trainer.py,
bin/run_collect_eval.py): gin parse -> entry point, exactly the
commands the research READMEs advertise."""

import glob
import os

from tensor2robot_amd.bin import run_collect_eval
from tensor2robot_amd.bin import run_t2r_trainer

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_run_t2r_trainer_main(tmp_path):
  config = os.path.join(
      REPO, "tensor2robot_amd", "research", "pose_env", "configs",
      "run_train_reg_model.gin")
  result = run_t2r_trainer.main([
      "--gin_configs", config,
      "--gin_bindings", f"train_eval_model.model_dir = '{tmp_path}'",
      "--gin_bindings", "train_eval_model.max_train_steps = 2",
      "--gin_bindings", "train_eval_model.eval_steps = 1",
  ])
  assert result["global_step"] == 2
  # A checkpoint landed in the model dir.
  assert glob.glob(os.path.join(str(tmp_path), "model.ckpt-*"))


def test_run_collect_eval_main(tmp_path):
  config = os.path.join(
      REPO, "tensor2robot_amd", "research", "pose_env", "configs",
      "run_random_collect.gin")
  run_collect_eval.main([
      "--gin_configs", config,
      "--gin_bindings", f"collect_eval_loop.root_dir = '{tmp_path}'",
      "--gin_bindings", "collect_eval_loop.max_loops = 1",
      "--gin_bindings", "run_meta_env.num_tasks = 1",
      "--gin_bindings", "run_meta_env.num_episodes_per_adaptation = 1",
  ])
  assert glob.glob(os.path.join(str(tmp_path), "policy_collect",
                                "*.tfrecord"))
