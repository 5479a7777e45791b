"""Record-batching input pipeline (the tf.data replacement).

Structure mirrors the reference's canonical input_fn
(`utils/tfdata.py:629-690` default_input_fn_tmpl): list_files(shuffle) ->
interleave(records) -> shuffle buffer -> repeat -> batch(drop_remainder)
-> parse (BATCH-before-parse, the reference's key perf property) ->
preprocess -> prefetch.

Prefetch is a background thread filling a bounded queue; on GPU the train
loop moves batches H2D on a dedicated copy stream with pinned staging.
"""

from __future__ import annotations

import itertools
import queue as queue_mod
import random
import threading
from typing import Callable, Dict, Iterator, List, Optional

import torch

from tensor2robot_amd.data import tfrecord
from tensor2robot_amd.specs import tensorspec_utils as tsu


class RecordBatchIterator:
  """Yields {dataset_key: [bytes]*batch} batches from TFRecord shards."""

  def __init__(self, dataset_map: Dict[str, List[str]], batch_size: int,
               shuffle: bool = True, repeat: bool = True,
               shuffle_buffer_size: int = 500, seed: Optional[int] = None,
               interleave_cycle: int = 4, shard_index: int = 0,
               num_shards: int = 1):
    self._dataset_map = dataset_map
    self._batch_size = batch_size
    self._shuffle = shuffle
    self._repeat = repeat
    self._buffer_size = shuffle_buffer_size
    self._rng = random.Random(seed)
    self._cycle = max(1, interleave_cycle)
    self._shard_index = shard_index
    self._num_shards = num_shards

  def _record_stream(self, files: List[str]) -> Iterator[bytes]:
    """Shard-shuffled interleaved record reading (tfdata.py:174-211)."""
    while True:
      order = list(files)
      if self._shuffle:
        self._rng.shuffle(order)
      if self._num_shards > 1:
        order = order[self._shard_index::self._num_shards] or order
      # Interleave `cycle` files at a time.
      for group_start in range(0, len(order), self._cycle):
        group = [tfrecord.read_records(p)
                 for p in order[group_start: group_start + self._cycle]]
        while group:
          alive = []
          for it in group:
            try:
              yield next(it)
              alive.append(it)
            except StopIteration:
              pass
          group = alive
      if not self._repeat:
        return

  def _shuffled(self, stream: Iterator[bytes]) -> Iterator[bytes]:
    if not self._shuffle:
      yield from stream
      return
    buf: List[bytes] = []
    for rec in stream:
      buf.append(rec)
      if len(buf) >= self._buffer_size:
        idx = self._rng.randrange(len(buf))
        buf[idx], buf[-1] = buf[-1], buf[idx]
        yield buf.pop()
    self._rng.shuffle(buf)
    yield from buf

  def __iter__(self):
    streams = {
        key: self._shuffled(self._record_stream(files))
        for key, files in self._dataset_map.items()
    }
    while True:
      batch: Dict[str, List[bytes]] = {}
      try:
        for key, stream in streams.items():
          batch[key] = [next(stream) for _ in range(self._batch_size)]
      except StopIteration:
        return  # drop_remainder=True
      yield batch


class WeightedRecordBatchIterator:
  """Samples each record from per-pattern streams with given weights.

  Reference `default_input_generator.py:229-301` WeightedRecordInputGenerator
  (used by BC-Z to mix 21-task/79-task data).
  """

  def __init__(self, file_patterns: List[str], weights: List[float],
               batch_size: int, seed: Optional[int] = None,
               shuffle_buffer_size: int = 500):
    self._iters = []
    for pattern in file_patterns:
      _, files = tfrecord.get_data_format_and_filenames(pattern)
      it = RecordBatchIterator({"": files}, batch_size=1, shuffle=True,
                               repeat=True,
                               shuffle_buffer_size=shuffle_buffer_size,
                               seed=seed)
      self._iters.append(iter(it._shuffled(it._record_stream(files))))
    total = sum(weights)
    self._weights = [w / total for w in weights]
    self._batch_size = batch_size
    self._rng = random.Random(seed)

  def __iter__(self):
    while True:
      batch = []
      for _ in range(self._batch_size):
        (stream,) = self._rng.choices(self._iters, weights=self._weights)
        batch.append(next(stream))
      yield {"": batch}


class PrefetchIterator:
  """Background-thread prefetch with a bounded queue (prefetch(AUTOTUNE))."""

  _SENTINEL = object()

  def __init__(self, source_fn: Callable[[], Iterator], depth: int = 4,
               pin_memory: bool = False):
    self._source_fn = source_fn
    self._depth = depth
    self._pin = pin_memory and torch.cuda.is_available()

  def __iter__(self):
    q: queue_mod.Queue = queue_mod.Queue(maxsize=self._depth)
    error = []

    def worker():
      try:
        for item in self._source_fn():
          if self._pin:
            item = _pin_struct(item)
          q.put(item)
      except BaseException as e:  # propagate to consumer
        error.append(e)
      finally:
        q.put(self._SENTINEL)

    t = threading.Thread(target=worker, daemon=True)
    t.start()
    while True:
      item = q.get()
      if item is self._SENTINEL:
        if error:
          raise error[0]
        return
      yield item


def _pin_struct(item):
  def pin(x):
    if isinstance(x, torch.Tensor) and not x.is_cuda:
      return x.pin_memory()
    return x
  if isinstance(item, tuple):
    return tuple(_pin_struct(x) for x in item)
  if isinstance(item, tsu.TensorSpecStruct):
    out = tsu.TensorSpecStruct()
    for k, v in item.items():
      out[k] = pin(v)
    return out
  if isinstance(item, dict):
    return {k: pin(v) for k, v in item.items()}
  return pin(item)


def move_struct_to_device(struct, device, non_blocking=True):
  """Moves every tensor in a (features, labels) struct to device."""
  if struct is None:
    return None
  if isinstance(struct, tuple):
    return tuple(move_struct_to_device(s, device, non_blocking)
                 for s in struct)
  if isinstance(struct, (tsu.TensorSpecStruct, dict)):
    out = tsu.TensorSpecStruct()
    for k, v in (struct.items() if hasattr(struct, "items") else []):
      out[k] = v.to(device, non_blocking=non_blocking) \
          if isinstance(v, torch.Tensor) else v
    return out
  if isinstance(struct, torch.Tensor):
    return struct.to(device, non_blocking=non_blocking)
  return struct
