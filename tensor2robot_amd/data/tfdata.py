"""Reference `utils/tfdata.py` public input API on the native pipeline.

The reference exposes a small set of entry points that user code builds
input pipelines from (`utils/tfdata.py:38-718`); this module maps each
name onto the MI355X-native machinery (data/tfrecord.py shard IO,
data/parser.py spec-driven parsing, data/pipeline.py batching):

  get_batch_size            :38-61   params-override resolution
  infer_data_format         :64-90   (re-exported from tfrecord.py)
  get_data_format_and_filenames[_list] :92-138
  get_dataset_metadata      :143-176 shard count + per-shard estimate
  serialized_to_parsed      :213-239 map the spec parser over batches
  default_input_fn_tmpl / get_input_fn :660-718

"datasets" here are plain python iterables of record batches
({dataset_key: [bytes]} or [bytes]) instead of tf.data objects.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

from tensor2robot_amd import gin
from tensor2robot_amd.data import parser as parser_mod
from tensor2robot_amd.data import pipeline
from tensor2robot_amd.data import tfrecord
from tensor2robot_amd.data.tfrecord import get_data_format_and_filenames
from tensor2robot_amd.data.tfrecord import infer_data_format  # noqa: F401
from tensor2robot_amd.utils import modes as run_modes

_log = logging.getLogger(__name__)


def get_batch_size(params: Optional[dict], batch_size: int) -> int:
  """Resolves the effective batch size (reference :38-61): an explicit
  `params['batch_size']` wins over the generator's."""
  params_batch_size = params.get("batch_size") if params else None
  if params_batch_size and params_batch_size != batch_size:
    _log.info(
        "The input_fn has a batch_size set through `params`, as well "
        "as in the input generator. These batch sizes do not match. "
        "Using the batch size %d from params", params_batch_size)
    return int(params_batch_size)
  return batch_size


def get_data_format_and_filenames_list(file_patterns: str):
  """(data_format, [files-per-pattern]) for comma-separated patterns
  (reference :92-117)."""
  import glob as _glob
  data_format = "tfrecord"
  filenames_list = []
  for p in file_patterns.split(","):
    data_format, rest = infer_data_format(p.strip())
    filenames_list.append(sorted(_glob.glob(rest)))
  if not any(filenames_list):
    raise ValueError(f"No files found for patterns {file_patterns}")
  return data_format, filenames_list


def get_dataset_metadata(file_patterns: str):
  """(data_format, num_shards, ~examples_per_shard) (reference :143-176);
  the estimate counts the records of the first shard."""
  data_format, files = get_data_format_and_filenames(file_patterns)
  num_shards = len(files)
  _log.info("Estimating dataset size from %s...", files[0])
  num_examples_per_shard = sum(1 for _ in tfrecord.read_records(files[0]))
  return data_format, num_shards, num_examples_per_shard


def serialized_to_parsed(dataset, feature_tspec, label_tspec,
                         num_parallel_calls: int = 2):
  """Maps the spec-driven parser over an iterable of record batches
  (reference :213-239).  Yields (features, labels) structs."""
  del num_parallel_calls  # the native parser releases the GIL per batch
  parse = parser_mod.create_parse_example_fn(feature_tspec, label_tspec)
  for batch in dataset:
    yield parse(batch)


@gin.configurable
def default_input_fn_tmpl(file_patterns: str, batch_size: int,
                          feature_spec, label_spec,
                          is_training: bool = True, preprocess_fn=None,
                          shuffle_buffer_size: int = 500,
                          seed: Optional[int] = None):
  """Record shards -> parsed (features, labels) batches (ref :660-689)."""
  _, files = get_data_format_and_filenames(file_patterns)
  records = pipeline.RecordBatchIterator(
      {"": files}, batch_size=batch_size, shuffle=is_training,
      repeat=is_training, shuffle_buffer_size=shuffle_buffer_size,
      seed=seed)
  for features, labels in serialized_to_parsed(records, feature_spec,
                                               label_spec):
    if preprocess_fn is not None:
      mode = run_modes.TRAIN if is_training else run_modes.EVAL
      features, labels = preprocess_fn(features, labels, mode)
    yield features, labels


def get_input_fn(feature_spec, label_spec, file_patterns: str, mode,
                 batch_size: int, preprocess_fn=None):
  """Returns input_fn(params) -> (features, labels) iterator
  (reference :692-718)."""

  def input_fn(params: Optional[dict] = None):
    used_batch_size = get_batch_size(params, batch_size)
    return default_input_fn_tmpl(
        file_patterns=file_patterns, batch_size=used_batch_size,
        feature_spec=feature_spec, label_spec=label_spec,
        is_training=(mode == run_modes.TRAIN),
        preprocess_fn=preprocess_fn)

  return input_fn
