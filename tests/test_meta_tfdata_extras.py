"""meta_tfdata.parallel_read and legacy meta_tf_models tests."""

import numpy as np
import pytest
import torch

from tensor2robot_amd.data import example as example_mod
from tensor2robot_amd.data import tfrecord as tfrecord_mod
from tensor2robot_amd.meta_learning import meta_tf_models
from tensor2robot_amd.meta_learning import meta_tfdata
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import mocks
from tensor2robot_amd.utils import modes as run_modes


def _write_task_file(path, task_id, n=12):
  writer = tfrecord_mod.TFRecordWriter(str(path))
  for i in range(n):
    writer.write(example_mod.encode_example(
        {"x": np.full(3, float(task_id), np.float32),
         "i": np.array([i], np.int64)}))
  writer.close()


def test_parallel_read_one_file_per_task(tmp_path):
  for t in range(3):
    _write_task_file(tmp_path / f"task{t}.tfrecord", t)

  def parse(records):
    decoded = [example_mod.decode_example(r) for r in records]
    return {"x": np.stack([d["x"] for d in decoded]),
            "i": np.stack([d["i"] for d in decoded])}

  it = meta_tfdata.parallel_read(
      str(tmp_path / "*.tfrecord"), parse, num_train_samples_per_task=2,
      num_val_samples_per_task=2, seed=0)
  batches = [next(it) for _ in range(6)]
  for b in batches:
    assert b["x"].shape == (4, 3)
    # All samples in a batch come from ONE task (one file).
    assert len(np.unique(b["x"])) == 1
  # All tasks appear within one epoch of 3 batches.
  tasks = {float(b["x"][0, 0]) for b in batches[:3]}
  assert tasks == {0.0, 1.0, 2.0}


def test_select_mode():
  train = tsu.TensorSpecStruct()
  train["a"] = torch.zeros(2, 3)
  val = tsu.TensorSpecStruct()
  val["a"] = torch.ones(2, 3)
  mode = torch.tensor([True, False])
  out = meta_tf_models.select_mode(mode, train, val)
  torch.testing.assert_close(out["a"][0], torch.ones(3))
  torch.testing.assert_close(out["a"][1], torch.zeros(3))


def test_create_meta_spec_and_preprocessor():
  base = mocks.MockT2RModel()
  spec = meta_tf_models._create_meta_spec(
      base.get_feature_specification(run_modes.TRAIN), "features", 4, 2)
  flat = tsu.flatten_spec_structure(spec)
  assert "train/measured_position" in flat
  assert "val/measured_position" in flat
  assert "val_mode" in flat
  assert tuple(flat["train/measured_position"].shape) == (4, 3)
  assert tuple(flat["val/measured_position"].shape) == (2, 3)

  prep = meta_tf_models.MetaPreprocessor(
      base_preprocessor=base.preprocessor,
      num_train_samples_per_task=4, num_val_samples_per_task=2)
  features = tsu.TensorSpecStruct()
  features["train/measured_position"] = torch.rand(3, 4, 3)
  features["val/measured_position"] = torch.rand(3, 2, 3)
  features["val_mode"] = torch.zeros(3, 1, dtype=torch.bool)
  f, _ = prep._preprocess_fn(features, None, run_modes.TRAIN)
  assert f["train/measured_position"].shape == (3, 4, 3)
  assert f["val_mode"].shape == (3, 1)


def test_metalearning_model_specs():
  base = mocks.MockT2RModel()

  class _RL2(meta_tf_models.MetalearningModel):

    def inference_network_fn(self, features, labels, mode, params=None):
      merged = meta_tf_models.select_mode(
          features["val_mode"],
          features["train"], features["val"])
      return {"x": merged["measured_position"]}

    def model_train_fn(self, features, labels, inference_outputs, mode,
                       params=None):
      return inference_outputs["x"].pow(2).mean()

  model = _RL2(
      base_model=base, device_type="cpu", compute_dtype="float32",
      num_train_samples_per_task=4, num_val_samples_per_task=2)
  spec = model.get_feature_specification(run_modes.TRAIN)
  assert "train/measured_position" in tsu.flatten_spec_structure(spec)
  assert isinstance(model.preprocessor, meta_tf_models.MetaPreprocessor)


def test_merge_expand_and_tile_val_mode():
  """Reference meta_tfdata tile_val_mode/merge_first_n_dims/
  expand_batch_dims parity."""
  import torch
  from tensor2robot_amd.meta_learning import meta_tfdata
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  s = tsu.TensorSpecStruct()
  s["a"] = torch.arange(24.0).reshape(2, 3, 4)
  merged = meta_tfdata.merge_first_n_dims(s, 2)
  assert merged["a"].shape == (6, 4)
  back = meta_tfdata.expand_batch_dims(merged, torch.tensor([2, 3]))
  torch.testing.assert_close(back["a"], s["a"])

  train = tsu.TensorSpecStruct(); train["x"] = torch.zeros(2, 3, 4)
  val = tsu.TensorSpecStruct(); val["x"] = torch.ones(2, 3, 4)
  pair = meta_tfdata.TrainValPair(train, val, torch.zeros(2, 1))
  tiled = meta_tfdata.tile_val_mode(pair)
  assert tiled.val_mode.shape == (6, 1)
  # Mismatched sample counts raise, like the reference.
  bad_val = tsu.TensorSpecStruct(); bad_val["x"] = torch.ones(2, 5, 4)
  import pytest
  with pytest.raises(ValueError):
    meta_tfdata.tile_val_mode(
        meta_tfdata.TrainValPair(train, bad_val, torch.zeros(2, 1)))


def test_pfor_map_fn_vmap():
  import torch
  from tensor2robot_amd.meta_learning import maml_model
  x = torch.randn(5, 3)
  out = maml_model.pfor_map_fn(lambda t: t * 2 + 1, x)
  torch.testing.assert_close(out, x * 2 + 1)


def test_meta_preprocessor_required_specs():
  """All legacy MetaPreprocessor in-specs are required (reference
  meta_tf_models_test.py:60-74): required-filtering is the identity."""
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  from tensor2robot_amd.utils import mocks
  from tensor2robot_amd.utils import modes as run_modes
  prep = meta_tf_models.MetaPreprocessor(
      base_preprocessor=mocks.MockT2RModel().preprocessor,
      num_train_samples_per_task=1, num_val_samples_per_task=1)
  for getter in (prep.get_in_feature_specification,
                 prep.get_in_label_specification):
    ref = tsu.flatten_spec_structure(getter(run_modes.TRAIN))
    filtered = tsu.filter_required_flat_tensor_spec(ref)
    assert dict(ref) == dict(filtered)
