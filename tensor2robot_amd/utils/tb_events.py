"""TensorBoard-compatible event file writer (no TF dependency).

Serializes `tensorflow.Event` protos with the native proto codec
(data/example.py wire helpers) into a `events.out.tfevents.*` TFRecord
so standard TensorBoard can read our training curves — the reference's
observability surface.

Wire schema (tensorflow/core/util/event.proto,
tensorflow/core/framework/summary.proto):
  Event:    wall_time=1 (double), step=2 (int64), file_version=3 (string),
            summary=5 (Summary)
  Summary:  repeated value=1
  Value:    tag=1 (string), simple_value=2 (float)
"""

from __future__ import annotations

import os
import socket
import struct
import time
from typing import Dict

from tensor2robot_amd.data import tfrecord as tfrecord_mod


def _varint(value: int) -> bytes:
  out = bytearray()
  while True:
    b = value & 0x7F
    value >>= 7
    if value:
      out.append(b | 0x80)
    else:
      out.append(b)
      return bytes(out)


def _tag(field: int, wire_type: int) -> bytes:
  return _varint((field << 3) | wire_type)


def _len_delim(field: int, payload: bytes) -> bytes:
  return _tag(field, 2) + _varint(len(payload)) + payload


def _double(field: int, v: float) -> bytes:
  return _tag(field, 1) + struct.pack("<d", v)


def _float(field: int, v: float) -> bytes:
  return _tag(field, 5) + struct.pack("<f", v)


def _int64(field: int, v: int) -> bytes:
  return _tag(field, 0) + _varint(v & 0xFFFFFFFFFFFFFFFF)


def encode_scalar_event(step: int, tag: str, value: float,
                        wall_time: float = None) -> bytes:
  summary_value = _len_delim(1, tag.encode("utf-8")) + _float(2, value)
  summary = _len_delim(1, summary_value)
  return (_double(1, wall_time if wall_time is not None else time.time())
          + _int64(2, int(step)) + _len_delim(5, summary))


def encode_file_version_event(wall_time: float = None) -> bytes:
  return (_double(1, wall_time if wall_time is not None else time.time())
          + _len_delim(3, b"brain.Event:2"))


class TBEventWriter:
  """Writes a TensorBoard-readable events.out.tfevents.* file."""

  def __init__(self, log_dir: str):
    os.makedirs(log_dir, exist_ok=True)
    name = f"events.out.tfevents.{int(time.time())}.{socket.gethostname()}"
    self._writer = tfrecord_mod.TFRecordWriter(
        os.path.join(log_dir, name))
    self._writer.write(encode_file_version_event())

  def add_scalar(self, tag: str, value: float, step: int):
    self._writer.write(encode_scalar_event(step, tag, float(value)))

  def add_scalars(self, scalars: Dict[str, float], step: int):
    for tag, value in scalars.items():
      self.add_scalar(tag, value, step)

  def flush(self):
    self._writer.flush()

  def close(self):
    self._writer.close()
