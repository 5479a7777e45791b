"""Predictor / policy / collect-eval-loop tests.

Mirrors the reference's 'distributed tests without a cluster' pattern
(SURVEY §4.8): train -> export -> predictor restore -> policy -> env loop,
all in-process through the filesystem contract.
"""

import os
import threading
import time

import numpy as np
import pytest
import torch

from tensor2robot_amd.data import input_generators
from tensor2robot_amd.models import optimizers
from tensor2robot_amd.policies import policies
from tensor2robot_amd.predictors import checkpoint_predictor
from tensor2robot_amd.predictors import ensemble_predictor
from tensor2robot_amd.predictors import exported_savedmodel_predictor as esp
from tensor2robot_amd.research.pose_env import pose_env
from tensor2robot_amd.research.pose_env import pose_env_models
from tensor2robot_amd.research.dql_grasping_lib import run_env as run_env_mod
from tensor2robot_amd.train import train_eval
from tensor2robot_amd.utils import continuous_collect_eval
from tensor2robot_amd.utils import cross_entropy
from tensor2robot_amd.utils import mocks
from tensor2robot_amd.utils import writer as writer_mod
from tensor2robot_amd.data import example as example_codec


def _train_and_export(tmp_path, steps=50):
  model = mocks.MockT2RModel(
      device_type="cpu",
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(5e-2))
  train_gen = mocks.MockInputGenerator(batch_size=16)
  eval_gen = mocks.MockInputGenerator(batch_size=16, seed=5)
  train_eval.train_eval_model(
      t2r_model=model, input_generator_train=train_gen,
      input_generator_eval=eval_gen, max_train_steps=steps, eval_steps=4,
      model_dir=str(tmp_path),
      create_exporters_fn=train_eval.create_default_exporters)
  export_dir = os.path.join(tmp_path, "export", "latest_exporter_numpy")
  return model, export_dir


def test_exported_predictor_roundtrip(tmp_path):
  model, export_dir = _train_and_export(tmp_path)
  predictor = esp.ExportedSavedModelPredictor(export_dir, timeout=5)
  assert predictor.is_loaded
  assert predictor.global_step == 50
  spec = predictor.get_feature_specification()
  assert "measured_position" in spec
  out = predictor.predict(
      {"measured_position": np.ones((3, 3), np.float32)})
  assert out["prediction"].shape == (3, 1)
  # Predictions are sign-consistent with the learned rule (sum > 0 -> 1).
  pos = predictor.predict(
      {"measured_position": np.full((2, 3), 0.9, np.float32)})
  neg = predictor.predict(
      {"measured_position": np.full((2, 3), -0.9, np.float32)})
  assert pos["prediction"].mean() > 0.5
  assert neg["prediction"].mean() < 0.5


def test_exported_predictor_times_out_gracefully(tmp_path):
  predictor = esp.ExportedSavedModelPredictor(
      str(tmp_path / "nothing"), timeout=0.5,
      restore_model_option=esp.RestoreOptions.DO_NOT_RESTORE)
  assert not predictor.restore()
  assert not predictor.is_loaded


def test_exported_predictor_async_restore(tmp_path):
  model, export_dir = _train_and_export(tmp_path, steps=10)
  predictor = esp.ExportedSavedModelPredictor(
      export_dir, timeout=10,
      restore_model_option=esp.RestoreOptions.RESTORE_ASYNCHRONOUSLY)
  assert predictor.is_loaded  # join guard waits for the thread


def test_checkpoint_predictor(tmp_path):
  model, _ = _train_and_export(tmp_path, steps=20)
  model2 = mocks.MockT2RModel(device_type="cpu")
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model2, checkpoint_dir=str(tmp_path), timeout=5)
  assert predictor.restore()
  assert predictor.global_step == 20
  out = predictor.predict(
      {"measured_position": np.ones((2, 3), np.float32)})
  assert "prediction" in out


def test_checkpoint_predictor_init_randomly():
  model = mocks.MockT2RModel(device_type="cpu")
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model, checkpoint_dir=None)
  predictor.init_randomly()
  out = predictor.predict(
      {"measured_position": np.zeros((1, 3), np.float32)})
  assert out["logit"].shape == (1, 1)


def test_ensemble_predictor(tmp_path):
  _, export_dir = _train_and_export(tmp_path, steps=10)
  predictor = ensemble_predictor.EnsembleExportedSavedModelPredictor(
      export_dirs=f"{export_dir},{export_dir}", ensemble_size=2,
      timeout=5, seed=0)
  assert predictor.restore()
  out = predictor.predict(
      {"measured_position": np.zeros((2, 3), np.float32)})
  assert out["prediction"].shape == (2, 1)


def test_cem_optimizer_finds_maximum():
  cem = cross_entropy.CrossEntropyMethod(num_samples=128, num_elites=12,
                                         num_iterations=10, seed=0)
  target = np.array([0.3, -0.7], np.float32)

  def objective(samples):
    return -np.sum((samples - target) ** 2, axis=1)

  best, score, mean, std = cem.run(objective, np.zeros(2, np.float32),
                                   np.ones(2, np.float32))
  assert np.linalg.norm(best - target) < 0.15


def test_regression_policy_env_loop(tmp_path):
  """Full predictor -> policy -> env -> replay-writer loop on pose_env."""
  model = pose_env_models.PoseEnvRegressionModel(
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-3))
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model, checkpoint_dir=None)
  predictor.init_randomly()
  policy = policies.RegressionPolicy(predictor=predictor,
                                     state_key="state/image")
  env = pose_env.PoseToyEnv(seed=1)
  replay_writer = writer_mod.TFRecordReplayWriter()

  def episode_to_transitions(episode_data):
    out = []
    for t in episode_data:
      out.append(example_codec.encode_example({
          "state/image": [b""],
          "action/pose": np.asarray(t.action, np.float32),
          "reward": np.asarray([t.reward], np.float32),
      }))
    return out

  rewards = run_env_mod.run_env(
      env, policy=policy, replay_writer=replay_writer,
      episode_to_transitions_fn=episode_to_transitions,
      root_dir=str(tmp_path), num_episodes=2, max_episode_steps=3)
  assert len(rewards) == 2
  collect_dir = os.path.join(tmp_path, "policy_collect")
  files = os.listdir(collect_dir)
  assert any(f.endswith(".tfrecord") for f in files)


def test_cem_policy_with_critic():
  # action_batch_size expands the serving action spec to [S, d] (CEM tiling).
  model = pose_env_models.PoseEnvContinuousMCModel(action_batch_size=16)
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model, checkpoint_dir=None)
  predictor.init_randomly()

  class PoseCEMPolicy(policies.CEMPolicy):

    def _split_action(self, action):
      return {"action/pose": action.astype(np.float32)}

  policy = PoseCEMPolicy(predictor=predictor, action_size=2,
                         cem_samples=16, cem_iterations=2,
                         state_key="state/image", seed=0)
  state = pose_env.PoseToyEnv(seed=3).reset()  # uint8 render (env contract)
  action = policy.SelectAction(state)
  assert action.shape == (2,)


def test_collect_eval_loop(tmp_path):
  """Actor loop polls exports produced by a trainer (filesystem contract)."""
  _, export_dir = _train_and_export(tmp_path, steps=10)

  calls = []

  class _StubPolicy:

    def restore(self):
      return True

    def init_randomly(self):
      pass

    @property
    def global_step(self):
      return 10 + len(calls)

    def reset(self):
      pass

    def sample_action(self, obs, explore_prob=0.0):
      return np.zeros(2, np.float32), {}

  def run_agent_fn(env, policy=None, global_step=0, root_dir="",
                   num_episodes=1, tag=""):
    calls.append((tag, global_step))

  result = continuous_collect_eval.collect_eval_loop(
      collect_env=object(), eval_env=object(),
      policy_class=_StubPolicy, run_agent_fn=run_agent_fn,
      root_dir=str(tmp_path), max_steps=10, poll_sleep_secs=0.01,
      max_loops=5)
  assert result >= 10
  assert ("collect", 10) in calls and ("eval", 10) in calls


def test_exploration_policies():
  model = pose_env_models.PoseEnvRegressionModel()
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model, checkpoint_dir=None)
  predictor.init_randomly()
  state = pose_env.PoseToyEnv(seed=4).reset()
  base = policies.RegressionPolicy(predictor=predictor,
                                   state_key="state/image")
  a0 = base.SelectAction(state)
  ou = policies.OUExploreRegressionPolicy(
      predictor=predictor, state_key="state/image", seed=0)
  ou.reset()
  a1 = ou.SelectAction(state)
  assert a1.shape == a0.shape and not np.allclose(a0, a1)
  sched = policies.ScheduledExplorationRegressionPolicy(
      predictor=predictor, state_key="state/image", seed=0)
  a2 = sched.SelectAction(state)
  assert a2.shape == a0.shape


def test_exported_predictor_serialized_examples(tmp_path):
  """tf_example receiver parity: feed serialized Examples to a servable."""
  import numpy as np
  from tensor2robot_amd.data import example as example_mod
  from tensor2robot_amd.export_generators import default_export_generator
  from tensor2robot_amd.utils import modes as run_modes

  model, export_dir = _train_and_export(tmp_path)
  predictor = esp.ExportedSavedModelPredictor(export_dir, timeout=5)
  assert predictor.is_loaded
  records = [example_mod.encode_example(
      {"measured_position": np.array([0.1, -0.2, 0.3], np.float32)})
      for _ in range(4)]
  out = predictor.predict_serialized(records)
  assert out["prediction"].shape[0] == 4


def test_grasping_cem_policy_cpu():
  """QT-Opt CEM serving path end-to-end on CPU (megabatch tiling)."""
  from tensor2robot_amd.research.qtopt import t2r_models
  model = t2r_models.GraspingModel(
      device_type="cpu", compute_dtype="float32", action_batch_size=8)
  predictor = checkpoint_predictor.CheckpointPredictor(t2r_model=model)
  predictor.init_randomly()
  policy = t2r_models.GraspingCEMPolicy(
      predictor=predictor, cem_samples=8, cem_iterations=1, seed=0)
  state = np.random.RandomState(0).randint(
      0, 256, (t2r_models.RAW_HEIGHT, t2r_models.RAW_WIDTH, 3)).astype(
          np.uint8)
  action = policy.SelectAction(state)
  assert action.shape == (t2r_models.ACTION_DIM,)


def test_warmup_requests_roundtrip(tmp_path):
  """Reference abstract_export_generator.py:109-142: zero-filled warmup
  request records a server replays before traffic."""
  from tensor2robot_amd.data import tfrecord
  from tensor2robot_amd.export_generators import default_export_generator
  model = mocks.MockT2RModel(device_type="cpu")
  gen = default_export_generator.DefaultExportGenerator()
  gen.set_specification_from_model(model)
  path = gen.create_warmup_requests_numpy([1, 4], str(tmp_path))
  assert os.path.basename(path) == "warmup_requests.tfrecord"
  records = list(tfrecord.read_records(path))
  # Batch-B request = B consecutive single-example records.
  assert len(records) == 1 + 4
  spec = gen.serving_input_spec()
  for rec in records:
    feats = example_codec.decode_example(rec)
    # Every required serving input appears, zero-filled, per example.
    for key, sp in spec.items():
      name = sp.name or key
      assert name in feats, (name, sorted(feats))
      import numpy as _np
      arr = _np.asarray(feats[name], dtype=_np.float64)
      assert float(_np.abs(arr).sum()) == 0.0


def test_tf_example_export_embeds_warmup_and_roundtrips(tmp_path):
  """a tf_example-mode export carries its parse
  contract (t2r_assets spec) + warmup requests INSIDE the artifact, and
  predict_serialized replays those records end to end."""
  from tensor2robot_amd.data import tfrecord
  from tensor2robot_amd.export_generators import default_export_generator
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  model = mocks.MockT2RModel(device_type="cpu")
  _ = model.network
  gen = default_export_generator.DefaultExportGenerator()
  gen.set_specification_from_model(model)
  export_dir = gen.export(model, str(tmp_path / "export"),
                          global_step=7, receiver_mode="tf_example",
                          warmup_batch_sizes=[1, 3])
  warmup_path = os.path.join(export_dir, tsu.EXTRA_ASSETS_DIRECTORY,
                             "warmup_requests.tfrecord")
  assert os.path.exists(warmup_path), "warmup requests not in artifact"

  predictor = esp.ExportedSavedModelPredictor(
      str(tmp_path / "export"), timeout=5)
  assert predictor.is_loaded
  records = list(tfrecord.read_records(warmup_path))
  assert len(records) == 1 + 3
  groups = [records[:1], records[1:]]
  for bs, group in zip((1, 3), groups):
    out = predictor.predict_serialized(group)
    for key, arr in out.items():
      assert arr.shape[0] == bs, (key, arr.shape)


def test_tf_example_receiver_matches_numpy_receiver(tmp_path):
  """The two receiver families produce the same feed for the same data
  (reference default_export_generator.py:42-133 parity)."""
  from tensor2robot_amd.data import example as example_mod
  from tensor2robot_amd.export_generators import default_export_generator

  model = mocks.MockT2RModel(device_type="cpu")
  gen = default_export_generator.DefaultExportGenerator()
  gen.set_specification_from_model(model)
  numpy_receiver = gen.create_serving_input_receiver_numpy_fn()
  example_receiver = gen.create_serving_input_receiver_tf_example_fn()

  arr = np.array([[0.5, -1.0, 2.0]], np.float32)
  spec = gen.serving_input_spec()
  key = list(spec.keys())[0]
  name = spec[key].name or key
  feed_np = numpy_receiver({key: arr})
  record = example_mod.encode_example({name: arr[0]})
  feed_ex = example_receiver([record])
  assert set(feed_np) == set(feed_ex)
  for k in feed_np:
    np.testing.assert_allclose(feed_np[k].numpy(),
                               np.asarray(feed_ex[k]), rtol=1e-6)


def test_generic_cross_entropy_method_list_and_dict():
  """Reference cross_entropy.py:30-107 API: list and dict batches."""
  np.random.seed(0)

  def sample_fn(mean, stddev):
    return list(mean + stddev * np.random.randn(64))

  def objective_fn(samples):
    return [-(s - 3.0) ** 2 for s in samples]

  def update_fn(params, elites):
    return {"mean": float(np.mean(elites)),
            "stddev": float(np.std(elites) + 1e-3)}

  samples, values, params = cross_entropy.cross_entropy_method(
      sample_fn, objective_fn, update_fn,
      {"mean": 0.0, "stddev": 2.0}, num_elites=8, num_iterations=10)
  assert abs(params["mean"] - 3.0) < 0.3
  assert len(samples) == 64 and len(values) == 64

  # Dict-batch form: keys sorted coherently by the same value order.
  def sample_fn_d(mean, stddev):
    xs = mean + stddev * np.random.randn(32)
    return {"x": list(xs), "tag": list(range(32))}

  def objective_fn_d(samples):
    return [-(x - 1.0) ** 2 for x in samples["x"]]

  def update_fn_d(params, elites):
    assert set(elites) == {"x", "tag"}
    assert len(elites["x"]) == 4
    return {"mean": float(np.mean(elites["x"])),
            "stddev": float(np.std(elites["x"]) + 1e-3)}

  _, _, params = cross_entropy.cross_entropy_method(
      sample_fn_d, objective_fn_d, update_fn_d,
      {"mean": 0.0, "stddev": 2.0}, num_elites=4, num_iterations=10)
  assert abs(params["mean"] - 1.0) < 0.4


def test_generic_cem_early_termination():
  calls = []

  def sample_fn(mean):
    calls.append(1)
    return [mean, mean + 1]

  _, _, _ = cross_entropy.cross_entropy_method(
      sample_fn, lambda s: [float(x) for x in s],
      lambda p, e: {"mean": max(e)}, {"mean": 5.0}, num_elites=1,
      num_iterations=50, threshold_to_terminate=2.0)
  assert len(calls) == 1  # first batch already exceeds the threshold


def test_normal_cross_entropy_method_converges():
  np.random.seed(1)

  def objective_fn(samples):
    return [-float(np.sum((s - 2.0) ** 2)) for s in samples]

  mean, stddev = cross_entropy.normal_cross_entropy_method(
      objective_fn, mean=[0.0, 0.0], stddev=[2.0, 2.0],
      num_samples=128, num_elites=16, num_iterations=12)
  assert np.allclose(mean, [2.0, 2.0], atol=0.3)
  assert np.all(np.asarray(stddev) < 1.0)


def test_exponential_decay_schedule():
  from tensor2robot_amd.utils import global_step_functions as gsf
  sched = gsf.exponential_decay(initial_value=1.0, decay_steps=100,
                                decay_rate=0.5, staircase=True)
  assert sched(0) == 1.0
  assert sched(99) == 1.0          # staircase: same interval
  assert sched(100) == 0.5
  assert sched(250) == 0.25
  smooth = gsf.exponential_decay(initial_value=1.0, decay_steps=100,
                                 decay_rate=0.5, staircase=False)
  assert abs(smooth(50) - 0.5 ** 0.5) < 1e-9


def test_full_lifecycle_train_export_collect_retrain(tmp_path):
  """The complete reference robotics loop in one test: train from the
  committed TFRecord fixture -> export a servable -> poll it with a
  predictor-backed policy -> collect episodes from the env into new
  TFRecords (run_env + TFRecordReplayWriter) -> retrain from the
  collected data.  (reference: utils/train_eval.py +
  export_generators + predictors + dql_grasping_lib/run_env +
  utils/writer composed end-to-end)."""
  import glob as globmod
  from tensor2robot_amd.research.dql_grasping_lib import run_env as re_mod
  from tensor2robot_amd.research.pose_env import episode_to_transitions
  from tensor2robot_amd.research.pose_env import pose_env
  from tensor2robot_amd.research.pose_env import pose_env_models
  from tensor2robot_amd.data import input_generators
  from tensor2robot_amd.utils import writer as writer_mod

  fixture = os.path.join(
      os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
      "test_data", "pose_env_test_data.tfrecord")

  # 1) Train from recorded data and export a servable.
  model = pose_env_models.PoseEnvRegressionModel(
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-3))
  train_dir = str(tmp_path / "train")
  train_eval.train_eval_model(
      t2r_model=model,
      input_generator_train=input_generators.DefaultRecordInputGenerator(
          file_patterns=fixture, batch_size=4, seed=1),
      input_generator_eval=input_generators.DefaultRecordInputGenerator(
          file_patterns=fixture, batch_size=4, seed=2),
      max_train_steps=5, eval_steps=2, model_dir=train_dir,
      create_exporters_fn=train_eval.create_default_exporters)
  export_dir = os.path.join(train_dir, "export",
                            "latest_exporter_numpy")

  # 2) Serve it through the polling predictor + a regression policy.
  predictor = esp.ExportedSavedModelPredictor(export_dir, timeout=5)
  assert predictor.restore()
  policy = policies.RegressionPolicy(predictor, state_key="state/image")
  assert policy.global_step == 5
  assert policy.model_path is not None

  # 3) Collect episodes into TFRecords.
  env = pose_env.PoseToyEnv(seed=3)
  collect_dir = str(tmp_path / "policy_collect")
  rewards = re_mod.run_env(
      env, policy=policy,
      episode_to_transitions_fn=(
          episode_to_transitions.episode_to_transitions_pose_toy),
      replay_writer=writer_mod.TFRecordReplayWriter(),
      root_dir=collect_dir, global_step=policy.global_step,
      num_episodes=3)
  assert len(rewards) == 3 and all(np.isfinite(r) for r in rewards)
  records = globmod.glob(os.path.join(collect_dir, "policy_collect",
                                      "*.tfrecord"))
  assert records, os.listdir(collect_dir)

  # 4) Retrain from the data the policy just collected.
  retrain_dir = str(tmp_path / "retrain")
  result = train_eval.train_eval_model(
      t2r_model=pose_env_models.PoseEnvRegressionModel(
          create_optimizer_fn=lambda:
          optimizers.create_adam_optimizer(1e-3)),
      input_generator_train=input_generators.DefaultRecordInputGenerator(
          file_patterns=os.path.join(collect_dir, "policy_collect",
                                     "*.tfrecord"),
          batch_size=2, seed=2),
      input_generator_eval=None, max_train_steps=3,
      model_dir=retrain_dir)
  assert result["global_step"] == 3
  assert np.isfinite(result["loss"])


def test_export_warmup_roundtrip_predict_serialized(tmp_path):
  """Export -> warmup-request records -> predict_serialized: the
  serialized-example serving contract end to end (reference
  default_export_generator.py:84-142)."""
  from tensor2robot_amd.data import tfrecord
  from tensor2robot_amd.export_generators import (
      abstract_export_generator)
  model, export_dir = _train_and_export(tmp_path, steps=10)

  gen = abstract_export_generator.AbstractExportGenerator()
  gen.set_specification_from_model(model)
  warmup_path = gen.create_warmup_requests_numpy([1, 3], str(tmp_path))
  records = list(tfrecord.read_records(warmup_path))
  assert len(records) == 4  # 1 + 3 single-example records

  predictor = esp.ExportedSavedModelPredictor(export_dir, timeout=5)
  assert predictor.restore()
  # Replay the batch-3 warmup group as one serialized predict call.
  out = predictor.predict_serialized(records[1:])
  assert out["prediction"].shape == (3, 1)
  # Serialized path == numpy path on the same (zero) input.
  ref = predictor.predict(
      {"measured_position": np.zeros((3, 3), np.float32)})
  np.testing.assert_allclose(out["prediction"], ref["prediction"],
                             rtol=1e-5)


def test_saved_model_v2_predictor_names(tmp_path):
  """The reference's three SavedModel predictor flavors resolve and
  serve here (saved_model_v2_predictor_test.py parity — one servable
  format, three names)."""
  from tensor2robot_amd.predictors import saved_model_v2_predictor as v2
  _, export_dir = _train_and_export(tmp_path, steps=10)
  for cls in (v2.SavedModelPredictorBase, v2.SavedModelTF1Predictor,
              v2.SavedModelTF2Predictor):
    predictor = cls(export_dir, timeout=5)
    assert predictor.restore()
    out = predictor.predict(
        {"measured_position": np.ones((2, 3), np.float32)})
    assert out["prediction"].shape == (2, 1)
