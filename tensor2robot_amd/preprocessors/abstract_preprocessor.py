"""Preprocessor contracts: validated in-graph per-batch transforms.

Reference: `preprocessors/abstract_preprocessor.py` (4 spec getters :87-133,
`preprocess` = validate_and_pack(in) -> _preprocess_fn -> validate_and_flatten
(out) :171-217), `noop_preprocessor.py`, `spec_transformation_preprocessor.py`.

In this framework preprocessors run on torch tensors — on CPU inside the
input pipeline or on GPU right after H2D transfer (the MI355X-native
equivalent of the reference's dataset.map stage; the heavy image ops are HIP
kernels in tensor2robot_amd/ops).
"""

from __future__ import annotations

import abc
from typing import Tuple

from tensor2robot_amd import gin
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes

TRAIN = run_modes.TRAIN
EVAL = run_modes.EVAL
PREDICT = run_modes.PREDICT


class AbstractPreprocessor(abc.ABC):
  """Validated transform between serialized data and model input."""

  def __init__(self, model_feature_specification_fn=None,
               model_label_specification_fn=None):
    for fn in (model_feature_specification_fn, model_label_specification_fn):
      if fn is not None:
        for mode in (TRAIN, EVAL, PREDICT):
          tsu.assert_valid_spec_structure(fn(mode))
    self._model_feature_specification_fn = model_feature_specification_fn
    self._model_label_specification_fn = model_label_specification_fn

  @property
  def model_feature_specification_fn(self):
    return self._model_feature_specification_fn

  @model_feature_specification_fn.setter
  def model_feature_specification_fn(self, fn):
    self._model_feature_specification_fn = fn

  @property
  def model_label_specification_fn(self):
    return self._model_label_specification_fn

  @model_label_specification_fn.setter
  def model_label_specification_fn(self, fn):
    self._model_label_specification_fn = fn

  # -- spec getters (reference :87-133) ------------------------------------
  @abc.abstractmethod
  def get_in_feature_specification(self, mode) -> tsu.TensorSpecStruct:
    """Specs of what the data pipeline must deliver to this preprocessor."""

  @abc.abstractmethod
  def get_in_label_specification(self, mode) -> tsu.TensorSpecStruct:
    pass

  @abc.abstractmethod
  def get_out_feature_specification(self, mode) -> tsu.TensorSpecStruct:
    """Specs of what this preprocessor delivers to the model."""

  @abc.abstractmethod
  def get_out_label_specification(self, mode) -> tsu.TensorSpecStruct:
    pass

  @abc.abstractmethod
  def _preprocess_fn(self, features, labels, mode):
    """The actual transform; operates on validated TensorSpecStructs."""

  def preprocess(self, features, labels, mode) -> Tuple:
    """validate_and_pack(in) -> _preprocess_fn -> validate_and_flatten(out)."""
    features = tsu.validate_and_pack(
        self.get_in_feature_specification(mode), features, ignore_batch=True)
    if labels is not None:
      labels = tsu.validate_and_pack(
          self.get_in_label_specification(mode), labels, ignore_batch=True)
    features, labels = self._preprocess_fn(features, labels, mode)
    features = tsu.validate_and_flatten(
        self.get_out_feature_specification(mode), features,
        ignore_batch=True)
    if labels is not None:
      labels = tsu.validate_and_flatten(
          self.get_out_label_specification(mode), labels, ignore_batch=True)
    return features, labels


@gin.configurable
class NoOpPreprocessor(AbstractPreprocessor):
  """Identity preprocessor: in == out == flattened model specs."""

  def get_in_feature_specification(self, mode):
    return tsu.flatten_spec_structure(
        self._model_feature_specification_fn(mode))

  def get_in_label_specification(self, mode):
    return tsu.flatten_spec_structure(
        self._model_label_specification_fn(mode))

  def get_out_feature_specification(self, mode):
    return tsu.flatten_spec_structure(
        self._model_feature_specification_fn(mode))

  def get_out_label_specification(self, mode):
    return tsu.flatten_spec_structure(
        self._model_label_specification_fn(mode))

  def _preprocess_fn(self, features, labels, mode):
    return features, labels


@gin.configurable
class SpecTransformationPreprocessor(NoOpPreprocessor):
  """Base for preprocessors whose in-specs are a transform of model specs.

  Subclasses override `_transform_in_feature_specification` (and the label
  variant) plus `_preprocess_fn` (reference
  `spec_transformation_preprocessor.py:88-146`).
  """

  def get_in_feature_specification(self, mode):
    return self._transform_in_feature_specification(
        tsu.flatten_spec_structure(
            self._model_feature_specification_fn(mode)), mode)

  def get_in_label_specification(self, mode):
    return self._transform_in_label_specification(
        tsu.flatten_spec_structure(
            self._model_label_specification_fn(mode)), mode)

  def _transform_in_feature_specification(self, flat_spec, mode):
    return flat_spec

  def _transform_in_label_specification(self, flat_spec, mode):
    return flat_spec


@gin.configurable
class DevicePreprocessorWrapper(AbstractPreprocessor):
  """bf16 device discipline (the reference's TPUPreprocessorWrapper analog).

  In-specs are the base preprocessor's with bf16 replaced by f32 (host-side
  parsing stays f32); out-specs drop optionals and cast f32->bf16 to halve
  the host->device traffic (reference `tpu_preprocessor_wrapper.py:75-157`).
  """

  def __init__(self, preprocessor: AbstractPreprocessor):
    super().__init__()
    self._preprocessor = preprocessor

  @property
  def preprocessor(self):
    return self._preprocessor

  @property
  def model_feature_specification_fn(self):
    return self._preprocessor.model_feature_specification_fn

  @property
  def model_label_specification_fn(self):
    return self._preprocessor.model_label_specification_fn

  def get_in_feature_specification(self, mode):
    import torch
    return tsu.replace_dtype(
        self._preprocessor.get_in_feature_specification(mode),
        torch.bfloat16, torch.float32)

  def get_in_label_specification(self, mode):
    import torch
    return tsu.replace_dtype(
        self._preprocessor.get_in_label_specification(mode),
        torch.bfloat16, torch.float32)

  def get_out_feature_specification(self, mode):
    return tsu.filter_required_flat_tensor_spec(
        self._preprocessor.get_out_feature_specification(mode))

  def get_out_label_specification(self, mode):
    return tsu.filter_required_flat_tensor_spec(
        self._preprocessor.get_out_label_specification(mode))

  def _preprocess_fn(self, features, labels, mode):
    in_f = tsu.cast_bfloat16_to_float32(features)
    in_l = tsu.cast_bfloat16_to_float32(labels) if labels is not None \
        else None
    out_f, out_l = self._preprocessor._preprocess_fn(in_f, in_l, mode)
    out_f = tsu.pack_flat_sequence_to_spec_structure(
        self.get_out_feature_specification(mode),
        tsu.cast_float32_to_bfloat16(
            out_f, self._preprocessor.get_out_feature_specification(mode)))
    if out_l is not None:
      out_l = tsu.pack_flat_sequence_to_spec_structure(
          self.get_out_label_specification(mode),
          tsu.cast_float32_to_bfloat16(
              out_l, self._preprocessor.get_out_label_specification(mode)))
    return out_f, out_l
