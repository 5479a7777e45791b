"""Input generators: spec-driven batch sources for training and eval.

Reference: `input_generators/abstract_input_generator.py` (:34-193) and
`input_generators/default_input_generator.py` (:48-301).  A generator holds
batch_size + the preprocessor's IN specs (pulled from the model via
`set_specification_from_model`) and yields (features, labels) structs with
the preprocessor already applied.
"""

from __future__ import annotations

import functools
import json
import os
from typing import Callable, Dict, Iterator, List, Optional

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.data import parser as parser_mod
from tensor2robot_amd.data import pipeline
from tensor2robot_amd.data import tfrecord
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes


class AbstractInputGenerator:
  """Holds batch size + specs; produces an iterator of (features, labels)."""

  def __init__(self, batch_size: int = 32):
    self._batch_size = batch_size
    self._feature_spec = None
    self._label_spec = None
    self._preprocess_fn = None
    self._out_feature_spec = None
    self._out_label_spec = None

  @property
  def batch_size(self) -> int:
    return self._batch_size

  @batch_size.setter
  def batch_size(self, value: int):
    self._batch_size = value

  def set_specification_from_model(self, model, mode):
    """Pulls the preprocessor IN-specs as the pipeline output contract.

    Reference `abstract_input_generator.py:76-98`.
    """
    preprocessor = model.preprocessor
    self._feature_spec = preprocessor.get_in_feature_specification(mode)
    self._label_spec = preprocessor.get_in_label_specification(mode)
    tsu.assert_valid_spec_structure(self._feature_spec)
    tsu.assert_valid_spec_structure(self._label_spec)
    self._out_feature_spec = preprocessor.get_out_feature_specification(mode)
    self._out_label_spec = preprocessor.get_out_label_specification(mode)
    self._preprocess_fn = functools.partial(preprocessor.preprocess,
                                            mode=mode)

  def set_feature_specifications(self, feature_spec, out_feature_spec=None):
    self._feature_spec = feature_spec
    self._out_feature_spec = out_feature_spec or feature_spec

  def set_label_specifications(self, label_spec, out_label_spec=None):
    self._label_spec = label_spec
    self._out_label_spec = out_label_spec or label_spec

  @property
  def feature_spec(self):
    return self._feature_spec

  @property
  def label_spec(self):
    return self._label_spec

  def defer_preprocessing(self):
    """Detach the bound preprocess_fn for on-device execution.

    After this, the pipeline yields RAW parsed batches and the caller
    (Trainer) applies the returned fn after H2D transfer — the deferred
    device-preprocess path (preprocess rides HIP, dynamic RNG stays
    outside the hipGraph).  Returns None if nothing is bound.
    """
    fn = self._preprocess_fn
    self._preprocess_fn = None
    return fn

  def set_preprocess_fn(self, preprocess_fn):
    """Mode must already be bound (reference :100-129 enforcement)."""
    if isinstance(preprocess_fn, functools.partial):
      if "mode" in (preprocess_fn.keywords or {}):
        self._preprocess_fn = preprocess_fn
        return
    if preprocess_fn is not None and \
        "mode" in getattr(preprocess_fn, "__code__", type(
            "x", (), {"co_varnames": ()})).co_varnames:
      raise ValueError(
          "preprocess_fn must have mode already filled in (use "
          "functools.partial(preprocess_fn, mode=...)).")
    self._preprocess_fn = preprocess_fn

  def create_dataset_input_fn(self, mode, prefetch_depth: int = 4,
                              pin_memory: bool = False,
                              h2d_device=None):
    """Returns a zero-arg callable producing the (features, labels) iterator."""
    run_modes.validate(mode)
    if self._feature_spec is None:
      raise ValueError(
          "set_specification_from_model must be called before "
          "create_dataset_input_fn.")

    # ONE PrefetchIterator per input_fn: its pinned staging ring must
    # survive re-iteration (the Trainer takes a fresh iterator per
    # train() segment; rebuilding the ring re-pays hipHostMalloc).
    prefetcher = pipeline.PrefetchIterator(
        lambda: self._iterate(mode), depth=prefetch_depth,
        pin_memory=pin_memory, h2d_device=h2d_device)

    def input_fn() -> Iterator:
      return iter(prefetcher)

    return input_fn

  def _apply_preprocess(self, features, labels):
    if self._preprocess_fn is None:
      return features, labels
    return self._preprocess_fn(features, labels)

  def _iterate(self, mode) -> Iterator:
    raise NotImplementedError


@gin.configurable
class DefaultRecordInputGenerator(AbstractInputGenerator):
  """TFRecord-file-backed generator (reference default_input_generator:48)."""

  def __init__(self, file_patterns: Optional[str] = None,
               dataset_map: Optional[Dict[str, str]] = None,
               batch_size: int = 32, shuffle_buffer_size: int = 500,
               seed: Optional[int] = None, shard_by_rank: bool = True):
    super().__init__(batch_size=batch_size)
    if bool(file_patterns) == bool(dataset_map):
      raise ValueError(
          "Exactly one of file_patterns or dataset_map must be set.")
    self._file_patterns = file_patterns
    self._dataset_map = dataset_map
    self._shuffle_buffer_size = shuffle_buffer_size
    self._seed = seed
    self._shard_by_rank = shard_by_rank

  def _resolve_files(self) -> Dict[str, List[str]]:
    if self._file_patterns:
      _, files = tfrecord.get_data_format_and_filenames(self._file_patterns)
      return {"": files}
    out = {}
    for key, patterns in self._dataset_map.items():
      _, files = tfrecord.get_data_format_and_filenames(patterns)
      out[key] = files
    return out

  def _shard_info(self):
    if self._shard_by_rank and "RANK" in os.environ and \
        "WORLD_SIZE" in os.environ:
      return int(os.environ["RANK"]), int(os.environ["WORLD_SIZE"])
    return 0, 1

  def _iterate(self, mode):
    parse = parser_mod.create_parse_example_fn(self._feature_spec,
                                               self._label_spec)
    shard_index, num_shards = self._shard_info()
    records = pipeline.RecordBatchIterator(
        self._resolve_files(), batch_size=self._batch_size,
        shuffle=(mode == run_modes.TRAIN),
        repeat=(mode == run_modes.TRAIN),
        shuffle_buffer_size=self._shuffle_buffer_size, seed=self._seed,
        shard_index=shard_index, num_shards=num_shards)
    for batch in records:
      features, labels = parse(batch)
      yield self._apply_preprocess(features, labels)


@gin.configurable
class FractionalRecordInputGenerator(DefaultRecordInputGenerator):
  """First `file_fraction` of files (data ablation; reference :105-126)."""

  def __init__(self, file_fraction: float = 1.0, **kwargs):
    super().__init__(**kwargs)
    self._file_fraction = file_fraction

  def _resolve_files(self):
    files = super()._resolve_files()
    if self._file_fraction < 1.0:
      for key in files:
        n = max(1, int(self._file_fraction * len(files[key])))
        files[key] = files[key][:n]
    return files


@gin.configurable
class MultiEvalRecordInputGenerator(DefaultRecordInputGenerator):
  """Selects the eval dataset by eval job name (reference :128-141)."""

  def __init__(self, eval_map: Dict[str, str] = None, **kwargs):
    self._eval_map = eval_map or {}
    multi_eval_name = get_multi_eval_name()
    if multi_eval_name and multi_eval_name in self._eval_map:
      kwargs["file_patterns"] = self._eval_map[multi_eval_name]
    elif self._eval_map and not kwargs.get("file_patterns"):
      kwargs["file_patterns"] = next(iter(self._eval_map.values()))
    super().__init__(**kwargs)


def get_multi_eval_name() -> Optional[str]:
  """Eval job naming from the cluster config env (reference :36-44)."""
  if "T2R_MULTI_EVAL_NAME" in os.environ:
    return os.environ["T2R_MULTI_EVAL_NAME"]
  tf_config = os.environ.get("TF_CONFIG")
  if tf_config:
    try:
      task = json.loads(tf_config).get("task", {})
      name = task.get("type", "")
      if name.startswith("eval_"):
        return name[len("eval_"):]
    except (ValueError, AttributeError):
      return None
  return None


@gin.configurable
class WeightedRecordInputGenerator(AbstractInputGenerator):
  """Per-file-pattern datasets sampled with weights (reference :229-301)."""

  def __init__(self, file_patterns: List[str], weights: List[float] = None,
               batch_size: int = 32, seed: Optional[int] = None,
               shuffle_buffer_size: int = 500):
    super().__init__(batch_size=batch_size)
    self._file_patterns = file_patterns
    self._weights = weights or [1.0] * len(file_patterns)
    if len(self._weights) != len(file_patterns):
      raise ValueError("weights and file_patterns length mismatch")
    self._seed = seed
    self._shuffle_buffer_size = shuffle_buffer_size

  def _iterate(self, mode):
    parse = parser_mod.create_parse_example_fn(self._feature_spec,
                                               self._label_spec)
    records = pipeline.WeightedRecordBatchIterator(
        self._file_patterns, self._weights, self._batch_size,
        seed=self._seed, shuffle_buffer_size=self._shuffle_buffer_size)
    for batch in records:
      features, labels = parse(batch)
      yield self._apply_preprocess(features, labels)


@gin.configurable
class GeneratorInputGenerator(AbstractInputGenerator):
  """Yields batches from a user generator of spec-conformant numpy structs.

  Reference :143-193; backbone of hermetic tests via the Random/Constant
  subclasses below.
  """

  def __init__(self, batch_size: int = 32, sequence_length: Optional[int] =
               None, max_batches: Optional[int] = None):
    super().__init__(batch_size=batch_size)
    self._sequence_length = sequence_length
    self._max_batches = max_batches

  def _generate_batch(self, batch_index: int):
    raise NotImplementedError

  def _iterate(self, mode):
    i = 0
    while self._max_batches is None or i < self._max_batches:
      features_np, labels_np = self._generate_batch(i)
      features = tsu.TensorSpecStruct()
      for k, v in tsu.flatten_spec_structure(features_np).items():
        features[k] = torch.as_tensor(v) if isinstance(v, np.ndarray) else v
      labels = None
      if labels_np is not None:
        labels = tsu.TensorSpecStruct()
        for k, v in tsu.flatten_spec_structure(labels_np).items():
          labels[k] = torch.as_tensor(v) if isinstance(v, np.ndarray) else v
      # Cast to declared spec dtypes (bf16 parsed as f32 host-side).
      features = _cast_to_spec(features, self._feature_spec)
      if labels is not None and self._label_spec is not None:
        labels = _cast_to_spec(labels, self._label_spec)
      yield self._apply_preprocess(features, labels)
      i += 1


def _cast_to_spec(struct, spec_structure):
  flat_spec = tsu.flatten_spec_structure(spec_structure)
  out = tsu.TensorSpecStruct()
  for k, v in struct.items():
    spec = flat_spec.get(k)
    if spec is not None and isinstance(v, torch.Tensor) and \
        v.dtype != spec.dtype and spec.dtype in (torch.bfloat16,
                                                 torch.float16):
      v = v.to(spec.dtype)
    out[k] = v
  return out


@gin.configurable
class DefaultRandomInputGenerator(GeneratorInputGenerator):
  """Spec-conformant random batches (reference :197-208)."""

  def __init__(self, seed: Optional[int] = None, **kwargs):
    super().__init__(**kwargs)
    self._seed = seed

  def _generate_batch(self, batch_index):
    seed = None if self._seed is None else self._seed + batch_index
    features = tsu.make_random_numpy(self._feature_spec, self._batch_size,
                                     self._sequence_length, seed=seed)
    labels = tsu.make_random_numpy(self._label_spec, self._batch_size,
                                   self._sequence_length, seed=seed) \
        if self._label_spec else None
    return features, labels


@gin.configurable
class DefaultConstantInputGenerator(GeneratorInputGenerator):
  """Spec-conformant constant batches (reference :210-226)."""

  def __init__(self, constant_value: float = 0.0, **kwargs):
    super().__init__(**kwargs)
    self._constant_value = constant_value

  def _generate_batch(self, batch_index):
    features = tsu.make_constant_numpy(self._feature_spec,
                                       self._constant_value,
                                       self._batch_size,
                                       self._sequence_length)
    labels = tsu.make_constant_numpy(self._label_spec, self._constant_value,
                                     self._batch_size,
                                     self._sequence_length) \
        if self._label_spec else None
    return features, labels
