"""TFRecord container IO (reader/writer) with CRC32C integrity.

Record layout (the on-disk format of the reference's datasets,
`utils/tfdata.py:29` TFRecordDataset):

  uint64 length | uint32 masked_crc32c(length) | bytes data |
  uint32 masked_crc32c(data)

masked_crc = ((crc >> 15) | (crc << 17)) + 0xa282ead8  (mod 2^32)

CRC32C (Castagnoli) is computed with a numpy-sliced table implementation;
the C++ extension provides the high-throughput path for training input.
"""

from __future__ import annotations

import glob as _glob
import os
import struct
from typing import Iterable, Iterator, List, Optional

import numpy as np

_CRC_TABLE = None


def _crc_table() -> np.ndarray:
  global _CRC_TABLE
  if _CRC_TABLE is None:
    poly = 0x82F63B78  # reflected CRC32C polynomial
    table = np.zeros(256, dtype=np.uint32)
    for i in range(256):
      crc = i
      for _ in range(8):
        crc = (crc >> 1) ^ (poly if crc & 1 else 0)
      table[i] = crc
    _CRC_TABLE = table
  return _CRC_TABLE


def crc32c(data: bytes) -> int:
  table = _crc_table()
  crc = np.uint32(0xFFFFFFFF)
  buf = np.frombuffer(data, dtype=np.uint8)
  # Python-loop over bytes is too slow for MB records; process in chunks via
  # table lookups with a small unrolled loop.  Still O(n) python-level ops,
  # so keep records modest on the pure-python path; C++ path for training.
  crc_val = 0xFFFFFFFF
  tab = table.tolist()
  for b in buf.tolist():
    crc_val = tab[(crc_val ^ b) & 0xFF] ^ (crc_val >> 8)
  return crc_val ^ 0xFFFFFFFF


def masked_crc32c(data: bytes) -> int:
  crc = crc32c(data)
  return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


class TFRecordWriter:
  """Writes TFRecord files (tmp-file + atomic rename on close)."""

  def __init__(self, path: str):
    self._path = path
    self._tmp_path = path + ".tmp"
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    self._file = open(self._tmp_path, "wb")

  def write(self, record: bytes):
    length = struct.pack("<Q", len(record))
    self._file.write(length)
    self._file.write(struct.pack("<I", masked_crc32c(length)))
    self._file.write(record)
    self._file.write(struct.pack("<I", masked_crc32c(record)))

  def flush(self):
    self._file.flush()

  def close(self):
    if self._file is not None:
      self._file.close()
      os.replace(self._tmp_path, self._path)
      self._file = None

  def __enter__(self):
    return self

  def __exit__(self, *exc):
    self.close()


def read_records(path: str, verify_crc: bool = False) -> Iterator[bytes]:
  """Iterates serialized records in one TFRecord file.

  Shards up to 256 MB read through the native reader
  (data/native/example_codec.cpp): framing + hardware-CRC32C
  verification run GIL-released in C++; larger shards stream through
  the python path below to bound memory."""
  try:
    from tensor2robot_amd.ops import _t2r_native
    native = _t2r_native.read_tfrecord_file
  except ImportError:
    native = None
  if native is not None and os.path.getsize(path) <= 256 * 1024 * 1024:
    try:
      yield from native(path, verify_crc)
    except RuntimeError as e:
      raise IOError(str(e)) from e
    return
  with open(path, "rb") as f:
    while True:
      header = f.read(12)
      if not header:
        return
      if len(header) < 12:
        raise IOError(f"Truncated TFRecord header in {path}")
      (length,) = struct.unpack("<Q", header[:8])
      if verify_crc:
        (expect,) = struct.unpack("<I", header[8:])
        actual = masked_crc32c(header[:8])
        if expect != actual:
          raise IOError(f"Corrupt length CRC in {path}")
      data = f.read(length)
      if len(data) < length:
        raise IOError(f"Truncated TFRecord data in {path}")
      footer = f.read(4)
      if verify_crc:
        (expect,) = struct.unpack("<I", footer)
        if expect != masked_crc32c(data):
          raise IOError(f"Corrupt data CRC in {path}")
      yield data


def list_files(file_patterns) -> List[str]:
  """Expands comma-separated glob patterns to a sorted file list."""
  if isinstance(file_patterns, str):
    file_patterns = file_patterns.split(",")
  out: List[str] = []
  for pattern in file_patterns:
    pattern = pattern.strip()
    if not pattern:
      continue
    matches = sorted(_glob.glob(pattern))
    if not matches and os.path.exists(pattern):
      matches = [pattern]
    out.extend(matches)
  return out


def infer_data_format(file_pattern: str):
  """Format prefix handling ('tfrecord:/path/*') — reference tfdata.py:64-90."""
  if ":" in file_pattern:
    prefix, rest = file_pattern.split(":", 1)
    if prefix in ("tfrecord",):
      return prefix, rest
    # windows-style or plain path with colon: treat whole as path
  return "tfrecord", file_pattern


def get_data_format_and_filenames(file_patterns):
  if isinstance(file_patterns, str):
    patterns = file_patterns.split(",")
  else:
    patterns = list(file_patterns)
  fmt = "tfrecord"
  paths = []
  for p in patterns:
    f, rest = infer_data_format(p.strip())
    fmt = f
    paths.append(rest)
  filenames = list_files(paths)
  if not filenames:
    raise ValueError(f"No files match {file_patterns!r}")
  return fmt, filenames
