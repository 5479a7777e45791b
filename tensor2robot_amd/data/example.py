"""tf.Example / tf.SequenceExample wire-format codec (pure Python).

Standalone implementation of the protobuf wire format for the Example family
of messages (the on-disk record format of the reference's data pipeline,
`utils/tfdata.py:273-541`).  No protobuf runtime required.

Wire schema (proto3 field numbers):
  Example        { Features features = 1; }
  Features       { map<string, Feature> feature = 1; }
  Feature        { BytesList bytes_list = 1; FloatList float_list = 2;
                   Int64List int64_list = 3; }
  BytesList      { repeated bytes value = 1; }
  FloatList      { repeated float value = 1 [packed]; }
  Int64List      { repeated int64 value = 1 [packed]; }
  SequenceExample{ Features context = 1; FeatureLists feature_lists = 2; }
  FeatureLists   { map<string, FeatureList> feature_list = 1; }
  FeatureList    { repeated Feature feature = 1; }

This is the CPU-reference codec; the production batch parser is the C++
extension in `tensor2robot_amd/cpp/example_parser.cc` (same semantics,
tested against this module).
"""

from __future__ import annotations

import struct
from typing import Dict, List, Sequence, Union

import numpy as np

FeatureValue = Union[List[bytes], np.ndarray]

_WT_VARINT = 0
_WT_I64 = 1
_WT_LEN = 2
_WT_I32 = 5


def _write_varint(out: bytearray, value: int):
  if value < 0:
    value += 1 << 64
  while True:
    b = value & 0x7F
    value >>= 7
    if value:
      out.append(b | 0x80)
    else:
      out.append(b)
      return


def _read_varint(data: bytes, pos: int):
  result = 0
  shift = 0
  while True:
    b = data[pos]
    pos += 1
    result |= (b & 0x7F) << shift
    if not b & 0x80:
      return result, pos
    shift += 7
    if shift >= 70:
      raise ValueError("Malformed varint")


def _write_tag(out: bytearray, field: int, wire_type: int):
  _write_varint(out, (field << 3) | wire_type)


def _write_len_delimited(out: bytearray, field: int, payload: bytes):
  _write_tag(out, field, _WT_LEN)
  _write_varint(out, len(payload))
  out.extend(payload)


# ---------------------------------------------------------------------------
# Feature encoding
# ---------------------------------------------------------------------------


def encode_feature(value) -> bytes:
  """Encodes one Feature message from bytes list / float array / int array."""
  out = bytearray()
  if isinstance(value, (bytes, str)):
    value = [value]
  if isinstance(value, (list, tuple)) and (
      not value or isinstance(value[0], (bytes, str))):
    inner = bytearray()
    for v in value:
      if isinstance(v, str):
        v = v.encode("utf-8")
      _write_len_delimited(inner, 1, v)
    _write_len_delimited(out, 1, bytes(inner))  # bytes_list
    return bytes(out)
  arr = np.asarray(value)
  if np.issubdtype(arr.dtype, np.floating):
    payload = arr.astype("<f4").ravel().tobytes()
    inner = bytearray()
    _write_tag(inner, 1, _WT_LEN)
    _write_varint(inner, len(payload))
    inner.extend(payload)
    _write_len_delimited(out, 2, bytes(inner))  # float_list (packed)
    return bytes(out)
  if np.issubdtype(arr.dtype, np.integer) or arr.dtype == np.bool_:
    inner = bytearray()
    varints = bytearray()
    for v in arr.astype(np.int64).ravel().tolist():
      _write_varint(varints, v)
    _write_tag(inner, 1, _WT_LEN)
    _write_varint(inner, len(varints))
    inner.extend(varints)
    _write_len_delimited(out, 3, bytes(inner))  # int64_list (packed)
    return bytes(out)
  raise ValueError(f"Unsupported feature value dtype {arr.dtype}")


def decode_feature(data: bytes):
  """Decodes one Feature message -> list[bytes] | float32 array | int64 array."""
  pos = 0
  while pos < len(data):
    tag, pos = _read_varint(data, pos)
    field, wt = tag >> 3, tag & 7
    if wt != _WT_LEN:
      raise ValueError(f"Unexpected wire type {wt} in Feature")
    ln, pos = _read_varint(data, pos)
    payload = data[pos: pos + ln]
    pos += ln
    if field == 1:
      return _decode_bytes_list(payload)
    if field == 2:
      return _decode_packed_floats(payload)
    if field == 3:
      return _decode_packed_int64(payload)
  # Empty feature.
  return []


def _decode_bytes_list(payload: bytes) -> List[bytes]:
  out = []
  pos = 0
  while pos < len(payload):
    tag, pos = _read_varint(payload, pos)
    if tag >> 3 != 1 or tag & 7 != _WT_LEN:
      raise ValueError("Malformed BytesList")
    ln, pos = _read_varint(payload, pos)
    out.append(payload[pos: pos + ln])
    pos += ln
  return out


def _decode_packed_floats(payload: bytes) -> np.ndarray:
  vals = []
  pos = 0
  while pos < len(payload):
    tag, pos = _read_varint(payload, pos)
    field, wt = tag >> 3, tag & 7
    if field != 1:
      raise ValueError("Malformed FloatList")
    if wt == _WT_LEN:  # packed
      ln, pos = _read_varint(payload, pos)
      vals.append(np.frombuffer(payload, dtype="<f4", count=ln // 4,
                                offset=pos))
      pos += ln
    elif wt == _WT_I32:  # unpacked single float
      vals.append(np.frombuffer(payload, dtype="<f4", count=1, offset=pos))
      pos += 4
    else:
      raise ValueError("Malformed FloatList wire type")
  if not vals:
    return np.zeros((0,), np.float32)
  return np.concatenate(vals).astype(np.float32, copy=False)


def _decode_packed_int64(payload: bytes) -> np.ndarray:
  vals = []
  pos = 0
  while pos < len(payload):
    tag, pos = _read_varint(payload, pos)
    field, wt = tag >> 3, tag & 7
    if field != 1:
      raise ValueError("Malformed Int64List")
    if wt == _WT_LEN:  # packed
      ln, pos = _read_varint(payload, pos)
      end = pos + ln
      while pos < end:
        v, pos = _read_varint(payload, pos)
        if v >= 1 << 63:
          v -= 1 << 64
        vals.append(v)
    elif wt == _WT_VARINT:
      v, pos = _read_varint(payload, pos)
      if v >= 1 << 63:
        v -= 1 << 64
      vals.append(v)
    else:
      raise ValueError("Malformed Int64List wire type")
  return np.asarray(vals, dtype=np.int64)


# ---------------------------------------------------------------------------
# Example / SequenceExample
# ---------------------------------------------------------------------------


def _encode_features(features: Dict[str, FeatureValue]) -> bytes:
  out = bytearray()
  for name, value in features.items():
    entry = bytearray()
    _write_len_delimited(entry, 1, name.encode("utf-8"))
    _write_len_delimited(entry, 2, encode_feature(value))
    _write_len_delimited(out, 1, bytes(entry))
  return bytes(out)


def encode_example(features: Dict[str, FeatureValue]) -> bytes:
  """Builds a serialized tf.Example from {name: value}."""
  out = bytearray()
  _write_len_delimited(out, 1, _encode_features(features))
  return bytes(out)


def _decode_features(data: bytes) -> Dict[str, object]:
  out = {}
  pos = 0
  while pos < len(data):
    tag, pos = _read_varint(data, pos)
    if tag >> 3 != 1 or tag & 7 != _WT_LEN:
      raise ValueError("Malformed Features")
    ln, pos = _read_varint(data, pos)
    entry = data[pos: pos + ln]
    pos += ln
    name, value = None, None
    epos = 0
    while epos < len(entry):
      etag, epos = _read_varint(entry, epos)
      eln, epos = _read_varint(entry, epos)
      payload = entry[epos: epos + eln]
      epos += eln
      if etag >> 3 == 1:
        name = payload.decode("utf-8")
      elif etag >> 3 == 2:
        value = payload
    if name is not None:
      out[name] = decode_feature(value or b"")
  return out


def decode_example(data: bytes) -> Dict[str, object]:
  """Parses a serialized tf.Example -> {name: list[bytes] | ndarray}."""
  pos = 0
  while pos < len(data):
    tag, pos = _read_varint(data, pos)
    if tag & 7 != _WT_LEN:
      raise ValueError("Malformed Example")
    ln, pos = _read_varint(data, pos)
    payload = data[pos: pos + ln]
    pos += ln
    if tag >> 3 == 1:
      return _decode_features(payload)
  return {}


def encode_sequence_example(
    context: Dict[str, FeatureValue],
    feature_lists: Dict[str, Sequence[FeatureValue]]) -> bytes:
  """Builds a serialized tf.SequenceExample."""
  out = bytearray()
  if context:
    _write_len_delimited(out, 1, _encode_features(context))
  fl_out = bytearray()
  for name, steps in feature_lists.items():
    fl = bytearray()
    for step in steps:
      _write_len_delimited(fl, 1, encode_feature(step))
    entry = bytearray()
    _write_len_delimited(entry, 1, name.encode("utf-8"))
    _write_len_delimited(entry, 2, bytes(fl))
    _write_len_delimited(fl_out, 1, bytes(entry))
  _write_len_delimited(out, 2, bytes(fl_out))
  return bytes(out)


def decode_sequence_example(data: bytes):
  """Parses tf.SequenceExample -> (context dict, {name: [per-step values]})."""
  context: Dict[str, object] = {}
  feature_lists: Dict[str, List[object]] = {}
  pos = 0
  while pos < len(data):
    tag, pos = _read_varint(data, pos)
    if tag & 7 != _WT_LEN:
      raise ValueError("Malformed SequenceExample")
    ln, pos = _read_varint(data, pos)
    payload = data[pos: pos + ln]
    pos += ln
    field = tag >> 3
    if field == 1:
      context = _decode_features(payload)
    elif field == 2:
      fpos = 0
      while fpos < len(payload):
        ftag, fpos = _read_varint(payload, fpos)
        fln, fpos = _read_varint(payload, fpos)
        entry = payload[fpos: fpos + fln]
        fpos += fln
        name, steps = None, []
        epos = 0
        while epos < len(entry):
          etag, epos = _read_varint(entry, epos)
          eln, epos = _read_varint(entry, epos)
          inner = entry[epos: epos + eln]
          epos += eln
          if etag >> 3 == 1:
            name = inner.decode("utf-8")
          elif etag >> 3 == 2:
            # FeatureList message: repeated Feature feature = 1.
            ipos = 0
            while ipos < len(inner):
              itag, ipos = _read_varint(inner, ipos)
              iln, ipos = _read_varint(inner, ipos)
              steps.append(decode_feature(inner[ipos: ipos + iln]))
              ipos += iln
        if name is not None:
          feature_lists[name] = steps
  return context, feature_lists
