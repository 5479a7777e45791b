"""End-to-end train/eval/export tests (reference train_eval_test.py shape)."""

import os

import numpy as np
import pytest
import torch

from tensor2robot_amd.models import optimizers
from tensor2robot_amd.train import checkpointing
from tensor2robot_amd.train import train_eval
from tensor2robot_amd.utils import mocks
from tensor2robot_amd.utils import modes as run_modes


def _make_model(**kwargs):
  kwargs.setdefault(
      "create_optimizer_fn",
      lambda: optimizers.create_adam_optimizer(learning_rate=5e-2))
  return mocks.MockT2RModel(device_type="cpu", **kwargs)


def _make_model_factory():
  def create_optimizer_fn():
    return optimizers.create_adam_optimizer(learning_rate=5e-2)
  return mocks.MockT2RModel(device_type="cpu",
                            create_optimizer_fn=create_optimizer_fn)


def test_train_converges_and_predictions_sign_correct(tmp_path):
  model = _make_model_factory()
  train_gen = mocks.MockInputGenerator(batch_size=32)
  eval_gen = mocks.MockInputGenerator(batch_size=32, seed=11)
  result = train_eval.train_eval_model(
      t2r_model=model,
      input_generator_train=train_gen,
      input_generator_eval=eval_gen,
      max_train_steps=300,
      eval_steps=10,
      model_dir=str(tmp_path),
      create_exporters_fn=train_eval.create_default_exporters,
      log_every_n_steps=100)
  assert result["global_step"] == 300
  assert result["eval_accuracy"] > 0.9
  # Output artifacts (reference assert_output_files).
  assert os.path.exists(os.path.join(tmp_path, "checkpoint"))
  assert checkpointing.latest_checkpoint(str(tmp_path)) is not None
  assert os.path.exists(os.path.join(tmp_path, "operative_config-0.gin"))
  assert os.path.exists(os.path.join(tmp_path, "events.jsonl"))
  export_root = os.path.join(tmp_path, "export", "latest_exporter_numpy")
  versions = [d for d in os.listdir(export_root) if d.isdigit()]
  assert versions, "latest exporter produced no export"
  export_dir = os.path.join(export_root, versions[-1])
  assert os.path.exists(os.path.join(export_dir, "servable.pt"))
  assert os.path.exists(
      os.path.join(export_dir, "assets.extra", "t2r_assets.pbtxt"))


def test_train_resume_restores_global_step(tmp_path):
  model = _make_model_factory()
  train_gen = mocks.MockInputGenerator(batch_size=8)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=train_gen,
      input_generator_eval=None, max_train_steps=20,
      model_dir=str(tmp_path))
  # Fresh model instance resumes from the checkpoint (reference :204-247).
  model2 = _make_model_factory()
  result = train_eval.train_eval_model(
      t2r_model=model2, input_generator_train=train_gen,
      input_generator_eval=None, max_train_steps=30,
      model_dir=str(tmp_path))
  assert result["global_step"] == 30
  ckpt = checkpointing.latest_checkpoint(str(tmp_path))
  assert checkpointing.global_step_from_path(ckpt) == 30


def test_eval_only_mode(tmp_path):
  model = _make_model_factory()
  train_gen = mocks.MockInputGenerator(batch_size=8)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=train_gen,
      input_generator_eval=None, max_train_steps=10,
      model_dir=str(tmp_path))
  model2 = _make_model_factory()
  eval_gen = mocks.MockInputGenerator(batch_size=8, seed=3)
  result = train_eval.train_eval_model(
      t2r_model=model2, input_generator_train=None,
      input_generator_eval=eval_gen, max_train_steps=10, eval_steps=5,
      model_dir=str(tmp_path))
  assert "accuracy" in result


def test_ema_swapping_checkpoint(tmp_path):
  model = _make_model(use_avg_model_params=True,
                      avg_model_params_decay=0.5)
  train_gen = mocks.MockInputGenerator(batch_size=8)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=train_gen,
      input_generator_eval=None, max_train_steps=5,
      model_dir=str(tmp_path))
  ckpt_path = checkpointing.latest_checkpoint(str(tmp_path))
  payload = torch.load(ckpt_path, map_location="cpu", weights_only=False)
  assert "ema_state" in payload and "raw_model_state" in payload
  # model_state holds AVERAGED weights, raw_model_state the live ones.
  name = next(iter(payload["ema_state"]["shadow"]))
  assert torch.allclose(payload["model_state"][name],
                        payload["ema_state"]["shadow"][name])


def test_predict_from_model(tmp_path):
  model = _make_model_factory()
  train_gen = mocks.MockInputGenerator(batch_size=16)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=train_gen,
      input_generator_eval=None, max_train_steps=100,
      model_dir=str(tmp_path))
  predict_gen = mocks.MockInputGenerator(batch_size=4, max_batches=1)
  preds = list(train_eval.predict_from_model(
      t2r_model=model, input_generator=predict_gen,
      model_dir=str(tmp_path)))
  assert len(preds) == 4
  assert "prediction" in preds[0]


def test_backup_checkpoint_for_eval(tmp_path):
  """Reference train_eval.py:616-684: copy-aside before eval, retry on
  in-flight tmp files, GC old backups."""
  model_dir = str(tmp_path)
  # No checkpoint yet -> gives up after retries.
  assert checkpointing.create_backup_checkpoint_for_eval(
      model_dir, max_retries=1, retry_sleep=0.01) is None
  for step in (3, 7):
    torch.save({"model_state": {}, "global_step": step},
               os.path.join(model_dir, f"model.ckpt-{step}.pt"))
  dst = checkpointing.create_backup_checkpoint_for_eval(
      model_dir, max_retries=2, retry_sleep=0.01)
  assert dst is not None and dst.endswith("model.ckpt-7.pt")
  assert os.path.exists(dst)
  # A newer checkpoint replaces the backup; the old backup is GC'd.
  torch.save({"model_state": {}, "global_step": 9},
             os.path.join(model_dir, "model.ckpt-9.pt"))
  dst2 = checkpointing.create_backup_checkpoint_for_eval(
      model_dir, max_retries=2, retry_sleep=0.01)
  assert dst2.endswith("model.ckpt-9.pt")
  backup_dir = os.path.dirname(dst2)
  assert sorted(os.listdir(backup_dir)) == ["model.ckpt-9.pt"]
  # An in-flight .tmp file blocks the copy until retries run out.
  open(os.path.join(model_dir, "model.ckpt-11.pt.tmp"), "w").close()
  torch.save({"model_state": {}, "global_step": 11},
             os.path.join(model_dir, "model.ckpt-11.pt"))
  assert checkpointing.create_backup_checkpoint_for_eval(
      model_dir, max_retries=1, retry_sleep=0.01) is None


def test_multi_eval_input_generator(tmp_path, monkeypatch):
  """Reference default_input_generator.py:128-141: the eval dataset is
  selected by the eval job name from the cluster env."""
  from tensor2robot_amd.data import input_generators
  monkeypatch.setenv("T2R_MULTI_EVAL_NAME", "holdout")
  gen = input_generators.MultiEvalRecordInputGenerator(
      eval_map={"train": "/data/a*.tfrecord",
                "holdout": "/data/b*.tfrecord"},
      batch_size=2)
  assert gen._file_patterns == "/data/b*.tfrecord"
  monkeypatch.delenv("T2R_MULTI_EVAL_NAME")
  # TF_CONFIG-style naming.
  monkeypatch.setenv(
      "TF_CONFIG", '{"task": {"type": "eval_train", "index": 0}}')
  gen2 = input_generators.MultiEvalRecordInputGenerator(
      eval_map={"train": "/data/a*.tfrecord",
                "holdout": "/data/b*.tfrecord"},
      batch_size=2)
  assert gen2._file_patterns == "/data/a*.tfrecord"


def test_reference_named_train_eval_helpers(tmp_path):
  """print_spec/print_specification/provide_input_generator.../save_copy
  (reference utils/train_eval.py:61-126,687-717)."""
  from tensor2robot_amd.train import train_eval as te
  model = mocks.MockT2RModel()
  te.print_specification(model)  # logs, must not raise
  gen = mocks.MockInputGenerator(batch_size=4)
  out = te.provide_input_generator_with_model_information(
      gen, model, run_modes.TRAIN)
  assert out is gen
  f, l = next(iter(gen.create_dataset_input_fn(run_modes.TRAIN)()))
  assert f["measured_position"].shape[0] == 4
  src = tmp_path / "src.txt"
  src.write_text("payload")
  dst = tmp_path / "dst.txt"
  assert te.save_copy(str(src), str(dst))
  assert dst.read_text() == "payload"
  assert not te.save_copy(str(src), str(dst))  # refuses overwrite
  assert te.save_copy(str(src), str(dst), overwrite=True)


def test_swapping_saver_writes_averaged_weights(tmp_path):
  """create_swapping_saver stores EMA weights as the canonical state
  (reference optimizers.py:150-160)."""
  import torch
  from tensor2robot_amd.models import optimizers
  net = torch.nn.Linear(3, 2)
  ema = optimizers.ExponentialMovingAverage(net, decay=0.5)
  with torch.no_grad():
    net.weight.add_(1.0)
  ema.update()
  saver = optimizers.create_swapping_saver(ema, str(tmp_path),
                                           max_to_keep=2)
  path = saver.save(5, net)
  state = torch.load(path, weights_only=False)
  assert torch.allclose(state["model_state"]["weight"],
                        ema.shadow["weight"])
  assert not torch.allclose(state["model_state"]["weight"], net.weight)


def test_training_is_seed_deterministic(tmp_path):
  """Same seeds -> bit-identical training trajectory on CPU (the
  torch-native analog of the reference's graph-level determinism)."""
  def run(seed, d):
    torch.manual_seed(seed)
    np.random.seed(seed)
    model = mocks.MockT2RModel(
        create_optimizer_fn=lambda: optimizers.create_adam_optimizer(
            1e-2))
    gen = mocks.MockInputGenerator(batch_size=8, seed=3)
    return train_eval.train_eval_model(
        t2r_model=model, input_generator_train=gen,
        input_generator_eval=None, max_train_steps=15,
        model_dir=str(d))["loss"]

  a = run(11, tmp_path / "a")
  b = run(11, tmp_path / "b")
  c = run(12, tmp_path / "c")
  assert a == b
  assert c != a


def test_lr_schedule_applies_per_step(tmp_path):
  """ScheduledOptimizer drives group['lr'] from the gin schedules
  (piecewise_linear / exponential_decay) during Trainer runs."""
  from tensor2robot_amd.utils import global_step_functions as gsf
  sched = gsf.piecewise_linear(boundaries=[0, 10], values=[1e-2, 1e-3])
  model = mocks.MockT2RModel(
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(
          learning_rate=sched))
  gen = mocks.MockInputGenerator(batch_size=4)
  gen.set_specification_from_model(model, run_modes.TRAIN)
  trainer = train_eval.Trainer(model, model_dir=str(tmp_path))
  trainer.train(gen.create_dataset_input_fn(run_modes.TRAIN), 10)
  final_lr = trainer.optimizer.optimizer.param_groups[0]["lr"]
  # Step 9 of the 0->10 ramp from 1e-2 to 1e-3.
  expected = sched(9)
  assert abs(final_lr - expected) < 1e-9, (final_lr, expected)
  assert final_lr < 1e-2
