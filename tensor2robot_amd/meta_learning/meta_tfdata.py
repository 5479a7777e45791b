"""Batch-reshaping utilities for meta-learning data.

Reference `meta_learning/meta_tfdata.py`: TrainValPair :27,
flatten_batch_examples :174 / unflatten_batch_examples :201
([tasks, samples, ...] <-> [tasks*samples, ...]), multi_batch_apply :261
(merge N batch dims -> f -> unmerge), split_train_val :130.
"""

from __future__ import annotations

import collections
from typing import Callable

import torch

from tensor2robot_amd.specs import tensorspec_utils as tsu

TrainValPair = collections.namedtuple("TrainValPair",
                                      ["train", "val", "val_mode"])
TrainValPair.__new__.__defaults__ = (None,)  # val_mode optional


def parallel_read(file_patterns, parse_fn, shuffle_filenames: bool = True,
                  num_train_samples_per_task: int = 4,
                  num_val_samples_per_task: int = 4,
                  shuffle_buffer_size: int = 50, filter_fn=None,
                  interleave_cycle_length=None, mode: str = "train",
                  seed=None):
  """ONE FILE == ONE TASK reading (reference meta_tfdata.py:32-126).

  Yields per-task batches: parse_fn applied to
  num_train + num_val consecutive (shuffled in train) records of each
  task file, interleaving tasks round-robin; task order reshuffles per
  epoch.  parse_fn: List[bytes] -> parsed batch struct.
  """
  import random as _random
  from tensor2robot_amd.data import tfrecord as tfrecord_mod

  _, filenames = tfrecord_mod.get_data_format_and_filenames(file_patterns)
  rng = _random.Random(seed)
  samples_per_task = (num_train_samples_per_task
                      + num_val_samples_per_task)

  def task_stream(path):
    """Infinite per-file batch stream (shuffled within a buffer)."""
    while True:
      records = list(tfrecord_mod.read_records(path))
      if mode == "train":
        rng.shuffle(records)
      for i in range(0, len(records) - samples_per_task + 1,
                     samples_per_task):
        batch = records[i: i + samples_per_task]
        parsed = parse_fn(batch)
        if filter_fn is not None and not filter_fn(parsed):
          continue
        yield parsed

  order = list(filenames)
  streams = {f: task_stream(f) for f in filenames}
  while True:
    if shuffle_filenames:
      rng.shuffle(order)
    for f in order:
      yield next(streams[f])


def _map_struct(fn, struct):
  if isinstance(struct, torch.Tensor):
    return fn(struct)
  if isinstance(struct, dict) or isinstance(struct, tsu.TensorSpecStruct):
    out = tsu.TensorSpecStruct()
    for k, v in tsu.flatten_spec_structure(struct).items():
      out[k] = fn(v) if isinstance(v, torch.Tensor) else v
    return out
  if isinstance(struct, (list, tuple)):
    return type(struct)(_map_struct(fn, s) for s in struct)
  return struct


def flatten_batch_examples(struct):
  """[tasks, samples, ...] -> [tasks*samples, ...] (reference :174)."""
  return _map_struct(lambda t: t.reshape(-1, *t.shape[2:]), struct)


def unflatten_batch_examples(struct, samples_per_task: int):
  """[tasks*samples, ...] -> [tasks, samples, ...] (reference :201)."""
  return _map_struct(
      lambda t: t.reshape(-1, samples_per_task, *t.shape[1:]), struct)


def multi_batch_apply(fn: Callable, num_batch_dims: int, *args, **kwargs):
  """Merge the first num_batch_dims dims, apply fn, unmerge (reference :261).

  All tensor arguments must share the leading batch dims.
  """
  batch_shape = None

  def find_shape(struct):
    nonlocal batch_shape
    if isinstance(struct, torch.Tensor) and batch_shape is None:
      batch_shape = struct.shape[:num_batch_dims]
    elif isinstance(struct, (dict, tsu.TensorSpecStruct)):
      for v in tsu.flatten_spec_structure(struct).values():
        find_shape(v)
    elif isinstance(struct, (list, tuple)):
      for v in struct:
        find_shape(v)

  for a in args:
    find_shape(a)
  if batch_shape is None:
    raise ValueError("multi_batch_apply found no tensors")

  def merge(t):
    return t.reshape(-1, *t.shape[num_batch_dims:])

  def unmerge(t):
    return t.reshape(*batch_shape, *t.shape[1:])

  merged_args = [_map_struct(merge, a) for a in args]
  merged_kwargs = {k: _map_struct(merge, v) for k, v in kwargs.items()}
  result = fn(*merged_args, **merged_kwargs)
  return _map_struct(unmerge, result)


def split_train_val(struct, num_train_samples: int) -> TrainValPair:
  """Split [tasks, train+val, ...] into a TrainValPair (reference :130)."""
  train = _map_struct(lambda t: t[:, :num_train_samples], struct)
  val = _map_struct(lambda t: t[:, num_train_samples:], struct)
  return TrainValPair(train, val)


def tile_val_mode(pair: TrainValPair) -> TrainValPair:
  """Tiles pair.val_mode over samples-per-task (reference :154-171).

  Requires num_train_samples_per_task == num_val_samples_per_task,
  like the reference; returns a new pair (namedtuples are immutable).
  """
  train_tensor = next(iter(
      tsu.flatten_spec_structure(pair.train).values()))
  val_tensor = next(iter(tsu.flatten_spec_structure(pair.val).values()))
  n_train, n_val = train_tensor.shape[1], val_tensor.shape[1]
  if n_train != n_val:
    raise ValueError("Flattening example and batch dimensions requires "
                     "num_train_samples and num_val_samples to be the "
                     "same.")
  if pair.val_mode is None:
    raise ValueError("pair.val_mode is not set")
  return pair._replace(val_mode=pair.val_mode.repeat(n_train, 1))


def merge_first_n_dims(structure, n: int):
  """Merges the first n dims of every tensor (reference :222-238)."""
  return _map_struct(lambda t: t.reshape(-1, *t.shape[n:]), structure)


def expand_batch_dims(structure, batch_sizes):
  """Unmerges dim 0 into `batch_sizes` (reference :241-257)."""
  sizes = [int(b) for b in batch_sizes]
  return _map_struct(lambda t: t.reshape(*sizes, *t.shape[1:]),
                     structure)
