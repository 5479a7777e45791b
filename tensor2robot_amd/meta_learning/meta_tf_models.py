"""Legacy RL^2-style meta model plumbing (TrainValPair specs).

Reference `meta_learning/meta_tf_models.py`: select_mode :51 (val_mode
switch between train/val tensors), _create_meta_spec :61 (train/ and
val/ prefixed copies + a bool val_mode spec), MetaPreprocessor :121
(base preprocessor applied to both splits on the flattened batch),
MetalearningModel :239 (base class for RL^2 models consuming
TrainValPairs).
"""

from __future__ import annotations

from typing import Optional

import torch

from tensor2robot_amd import gin
from tensor2robot_amd.meta_learning import meta_tfdata
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.preprocessors import abstract_preprocessor
from tensor2robot_amd.specs import tensorspec_utils as tsu

TSPEC = tsu.ExtendedTensorSpec


def select_mode(val_mode, train, val):
  """Elementwise where(val_mode, val, train) over structures (ref :51)."""
  train_flat = tsu.flatten_spec_structure(train)
  val_flat = tsu.flatten_spec_structure(val)
  out = tsu.TensorSpecStruct()
  for key in train_flat.keys():
    t, v = train_flat[key], val_flat[key]
    mask = val_mode
    if isinstance(mask, torch.Tensor):
      while mask.dim() < t.dim():
        mask = mask.unsqueeze(-1)
      out[key] = torch.where(mask.bool(), v, t)
    else:
      out[key] = v if mask else t
  return out


def _create_meta_spec(tensor_spec, spec_type: str,
                      num_train_samples_per_task: int,
                      num_val_samples_per_task: int):
  """TrainValPair spec with val_mode switch (reference :61-118)."""
  if spec_type not in ("features", "labels"):
    raise ValueError('We only support spec_type "features" or "labels" '
                     f"but received {spec_type}.")
  out = tsu.TensorSpecStruct()
  train_spec = tsu.copy_tensorspec(tensor_spec,
                                   batch_size=num_train_samples_per_task,
                                   prefix="train")
  for key, value in tsu.flatten_spec_structure(train_spec).items():
    out["train/" + key] = TSPEC.from_spec(value, is_optional=False)
  val_spec = tsu.copy_tensorspec(tensor_spec,
                                 batch_size=num_val_samples_per_task,
                                 prefix="val")
  for key, value in tsu.flatten_spec_structure(val_spec).items():
    out["val/" + key] = TSPEC.from_spec(value, is_optional=False)
  out["val_mode"] = TSPEC((1,), torch.bool,
                          name=f"val_mode/{spec_type}")
  return out


@gin.configurable
class MetaPreprocessor(abstract_preprocessor.AbstractPreprocessor):
  """Wraps a base preprocessor into TrainValPairs (reference :121-237)."""

  def __init__(self, base_preprocessor=None,
               num_train_samples_per_task: int = 4,
               num_val_samples_per_task: int = 4, **kwargs):
    super().__init__(**kwargs)
    self._base_preprocessor = base_preprocessor
    self._num_train = num_train_samples_per_task
    self._num_val = num_val_samples_per_task

  @property
  def num_train_samples_per_task(self):
    return self._num_train

  @property
  def num_val_samples_per_task(self):
    return self._num_val

  @property
  def base_preprocessor(self):
    return self._base_preprocessor

  def get_in_feature_specification(self, mode):
    return _create_meta_spec(
        self._base_preprocessor.get_in_feature_specification(mode),
        "features", self._num_train, self._num_val)

  def get_in_label_specification(self, mode):
    return _create_meta_spec(
        self._base_preprocessor.get_in_label_specification(mode),
        "labels", self._num_train, self._num_val)

  def get_out_feature_specification(self, mode):
    return _create_meta_spec(
        self._base_preprocessor.get_out_feature_specification(mode),
        "features", self._num_train, self._num_val)

  def get_out_label_specification(self, mode):
    return _create_meta_spec(
        self._base_preprocessor.get_out_label_specification(mode),
        "labels", self._num_train, self._num_val)

  def _split(self, struct, prefix):
    out = tsu.TensorSpecStruct()
    for k, v in tsu.flatten_spec_structure(struct).items():
      if k.startswith(prefix + "/"):
        out[k[len(prefix) + 1:]] = v
    return out

  def _preprocess_fn(self, features, labels, mode):
    if mode is None:
      raise ValueError("The mode should never be None.")
    out_f = tsu.TensorSpecStruct()
    out_l = tsu.TensorSpecStruct() if labels is not None else None
    for prefix, n in (("train", self._num_train), ("val", self._num_val)):
      f = meta_tfdata.flatten_batch_examples(self._split(features, prefix))
      l = meta_tfdata.flatten_batch_examples(self._split(labels, prefix)) \
          if labels is not None else None
      f, l = self._base_preprocessor._preprocess_fn(f, l, mode)
      for k, v in meta_tfdata.unflatten_batch_examples(f, n).items():
        out_f[f"{prefix}/{k}"] = v
      if out_l is not None and l is not None:
        for k, v in meta_tfdata.unflatten_batch_examples(l, n).items():
          out_l[f"{prefix}/{k}"] = v
    out_f["val_mode"] = features["val_mode"].reshape(-1, 1)
    if out_l is not None and "val_mode" in labels:
      out_l["val_mode"] = labels["val_mode"].reshape(-1, 1)
    return out_f, out_l


@gin.configurable
class MetalearningModel(abstract_model.AbstractT2RModel):
  """Base class for RL^2-style models over TrainValPairs (reference :239).

  Subclasses implement inference_network_fn consuming features with
  train/, val/ splits and a val_mode switch (use select_mode).
  """

  def __init__(self, base_model=None, preprocessor_cls=None,
               num_train_samples_per_task: int = 4,
               num_val_samples_per_task: int = 4, **kwargs):
    super().__init__(**kwargs)
    self._base_model = base_model
    self._meta_preprocessor_cls = preprocessor_cls or MetaPreprocessor
    self._num_train = num_train_samples_per_task
    self._num_val = num_val_samples_per_task

  @property
  def base_model(self):
    return self._base_model

  @property
  def preprocessor(self):
    if self._preprocessor is None:
      self._preprocessor = self._meta_preprocessor_cls(
          base_preprocessor=self._base_model.preprocessor,
          num_train_samples_per_task=self._num_train,
          num_val_samples_per_task=self._num_val)
    return self._preprocessor

  def get_feature_specification(self, mode):
    return _create_meta_spec(
        self._base_model.get_feature_specification(mode), "features",
        self._num_train, self._num_val)

  def get_label_specification(self, mode):
    return _create_meta_spec(
        self._base_model.get_label_specification(mode), "labels",
        self._num_train, self._num_val)

  def create_network(self):
    return self._base_model.network
