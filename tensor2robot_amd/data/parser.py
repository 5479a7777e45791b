"""Spec-driven batch parser: serialized Examples -> validated tensor structs.

The centerpiece of the data plumbing (reference `utils/tfdata.py:273-541`
`create_parse_tf_example_fn`): from feature/label specs the framework
auto-generates a parser mapping BATCHES of serialized tf.Example /
tf.SequenceExample protos to validated TensorSpecStructs of torch tensors.

Behavior carried over from the reference:
  * batching happens BEFORE parsing (the parse fn receives a batch).
  * multi-dataset routing via spec.dataset_key (:241-271).
  * bf16-declared specs are parsed as f32, cast at the end (:326-391).
  * sequence specs parse from the SequenceExample feature_lists half and get
    a companion '<key>_length' int64 tensor (:352-383).
  * encoded-image specs (JPEG/PNG) decode to uint8/uint16, zero image on
    empty string (:426-484).
  * varlen specs pad (varlen_default_value) or clip to shape[0] (:508-513).
  * final validate_and_pack into features/labels (:515-540).

This python implementation is the reference semantics; the C++ extension
(`_t2r_native.parse_example_batch`) accelerates the wire parsing and is used
automatically when built.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from tensor2robot_amd.data import example as example_codec
from tensor2robot_amd.data import image_codec
from tensor2robot_amd.specs import tensorspec_utils as tsu


def _np_parse_dtype(spec: tsu.ExtendedTensorSpec) -> np.dtype:
  """Numpy dtype the wire value is materialized as (bf16 -> f32)."""
  if spec.dtype in (torch.bfloat16, torch.float16):
    return np.dtype(np.float32)
  return spec.np_dtype


def _wire_category(spec: tsu.ExtendedTensorSpec) -> str:
  if tsu.is_encoded_image_spec(spec):
    return "bytes"
  np_dt = _np_parse_dtype(spec)
  if np.issubdtype(np_dt, np.floating):
    return "float"
  return "int64"


def _feature_to_array(value, spec, key) -> np.ndarray:
  """Normalizes one decoded Feature to the spec's flat numpy form."""
  cat = _wire_category(spec)
  if cat == "bytes":
    if isinstance(value, np.ndarray):
      raise ValueError(f"Feature {key} expected bytes, got numeric")
    return value  # list[bytes]
  if isinstance(value, list):
    value = np.asarray(value)
  np_dt = _np_parse_dtype(spec)
  return value.astype(np_dt, copy=False)


def _decode_image_batch(bytes_list: List[bytes],
                        spec: tsu.ExtendedTensorSpec) -> np.ndarray:
  if len(spec.shape) < 3:
    raise ValueError(
        f"Image spec {spec.name} must be >=3D (h,w,c), got {spec.shape}")
  single_dims = tuple(spec.shape[-3:])
  if single_dims[2] not in (1, 3):
    raise ValueError(
        f"Image spec {spec.name} channels must be 1 or 3: {spec.shape}")
  if spec.dtype not in (torch.uint8, torch.int32):
    raise ValueError(
        f"Encoded image spec {spec.name} must be uint8/uint16 dtype")
  np_dt = spec.np_dtype
  out = np.zeros((len(bytes_list),) + single_dims, dtype=np_dt)

  def decode_one(i: int, raw: bytes):
    if not raw:
      return  # zero image on empty string (reference :465-473)
    img = image_codec.decode_image(raw, spec.data_format)
    if img.ndim == 2:
      img = img[:, :, None]
    if img.shape[2] != single_dims[2]:
      if single_dims[2] == 1:
        img = img[:, :, :1]
      elif single_dims[2] == 3 and img.shape[2] == 1:
        img = np.repeat(img, 3, axis=2)
      else:
        img = img[:, :, :3]
    if img.shape[:2] != single_dims[:2]:
      raise ValueError(
          f"Image for {spec.name} has shape {img.shape}, spec wants "
          f"{single_dims}")
    out[i] = img.astype(np_dt, copy=False)

  # JPEG decode is ~1 ms+/image of GIL-released native work — batches
  # parallelize across host threads (same policy as
  # gpu_jpeg._huffman_batch).  PNGs stay serial: zlib + the native
  # unfilter run ~0.2 ms/image and the python glue dominates, so the
  # pool only adds contention there (measured 9.4 -> 13.2 ms/batch64).
  is_jpeg = (spec.data_format or "").lower() in ("jpeg", "jpg")
  if is_jpeg and len(bytes_list) >= 4:
    list(_decode_pool().map(lambda iv: decode_one(*iv),
                            enumerate(bytes_list)))
  else:
    for i, raw in enumerate(bytes_list):
      decode_one(i, raw)
  return out


_DECODE_POOL = None


def _decode_pool():
  global _DECODE_POOL
  if _DECODE_POOL is None:
    import concurrent.futures
    import os as _os
    _DECODE_POOL = concurrent.futures.ThreadPoolExecutor(
        min(16, _os.cpu_count() or 4))
  return _DECODE_POOL


def _spec_elements(spec: tsu.ExtendedTensorSpec) -> int:
  n = 1
  for d in spec.shape:
    if d is not None:
      n *= d
  return n


class ExampleParser:
  """Parses batches of serialized records for one dataset_key's spec set."""

  def __init__(self, specs: tsu.TensorSpecStruct, decode_images: bool = True,
               image_decode_device: str = ""):
    # image_decode_device="cuda": JPEG batches decode through the GPU
    # path (data/gpu_jpeg.py — host-thread Huffman + HIP idct/color),
    # yielding a CUDA uint8 tensor directly; 15x the CPU codec
    # (profiles/r2_jpeg_gpu.md).
    self._specs = specs
    self._decode_images = decode_images
    self._image_decode_device = image_decode_device
    self._has_sequence = any(
        s.is_sequence for s in specs.values())

  @staticmethod
  def _decode_example_batch(records: List[bytes]) -> List[Dict]:
    """Whole-batch wire decode: the C++ scanner runs GIL-released
    (data/native/example_codec.cpp) when built, else the python codec."""
    try:
      from tensor2robot_amd.ops import _t2r_native
      return _t2r_native.parse_example_batch(list(records))
    except ImportError:
      return [example_codec.decode_example(r) for r in records]

  def __call__(self, records: List[bytes]) -> tsu.TensorSpecStruct:
    """records: batch of serialized Example/SequenceExample protos."""
    batch = len(records)
    per_key_values: Dict[str, list] = {k: [] for k in self._specs.keys()}
    lengths: Dict[str, List[int]] = {}

    if self._has_sequence:
      try:
        from tensor2robot_amd.ops import _t2r_native
        decoded = _t2r_native.parse_sequence_example_batch(list(records))
      except ImportError:
        decoded = [example_codec.decode_sequence_example(r)
                   for r in records]
    else:
      decoded = [(c, {}) for c in self._decode_example_batch(records)]
    for context, feature_lists in decoded:
      for key, spec in self._specs.items():
        name = spec.name or key
        if spec.is_sequence:
          if name not in feature_lists:
            raise ValueError(
                f"Record missing sequence feature {name!r}; has "
                f"{list(feature_lists.keys())}")
          steps = [
              _feature_to_array(v, spec, key) for v in feature_lists[name]]
          per_key_values[key].append(steps)
          lengths.setdefault(key, []).append(len(steps))
        else:
          if name not in context:
            if spec.is_optional:
              per_key_values[key].append(None)
              continue
            raise ValueError(
                f"Record missing feature {name!r}; has "
                f"{list(context.keys())}")
          per_key_values[key].append(
              _feature_to_array(context[name], spec, key))

    out = tsu.TensorSpecStruct()
    for key, spec in self._specs.items():
      values = per_key_values[key]
      if all(v is None for v in values) and spec.is_optional:
        continue
      if spec.is_sequence:
        out[key] = self._assemble_sequence(key, spec, values)
        out[key + "_length"] = torch.as_tensor(lengths[key],
                                               dtype=torch.int64)
      else:
        out[key] = self._assemble_context(key, spec, values)
    return out

  # -- assembly ------------------------------------------------------------
  def _assemble_context(self, key, spec, values) -> torch.Tensor:
    if tsu.is_encoded_image_spec(spec) and self._decode_images:
      flat_bytes = []
      for v in values:
        if isinstance(v, list) and v and isinstance(v[0], (bytes, str)):
          flat_bytes.append(v[0] if v else b"")
        elif isinstance(v, list) and not v:
          flat_bytes.append(b"")
        else:
          raise ValueError(f"Image feature {key} is not bytes")
      if self._image_decode_device and \
          (spec.data_format or "").lower() in ("jpeg", "jpg") and \
          all(flat_bytes):
        from tensor2robot_amd.data import gpu_jpeg
        dec = gpu_jpeg.decode_jpeg_batch(flat_bytes,
                                         self._image_decode_device)
        want_c = spec.shape[-1]
        if dec.shape[-1] != want_c:
          dec = dec[..., :1] if want_c == 1 else \
              dec.expand(-1, -1, -1, 3).contiguous()
        return dec
      decoded = _decode_image_batch(flat_bytes, spec)
      return torch.from_numpy(decoded)
    if tsu.is_encoded_image_spec(spec):
      raise ValueError(
          f"decode_images=False unsupported for tensor output of {key}")
    if spec.varlen_default_value is not None:
      rows = []
      inner = _spec_elements(
          tsu.ExtendedTensorSpec(spec.shape[1:], spec.dtype)) \
          if len(spec.shape) > 1 else 1
      for v in values:
        arr = np.asarray(v)
        n = arr.size // inner if inner else 0
        arr = arr.reshape((n,) + tuple(spec.shape[1:]))
        arr = tsu.pad_or_clip_tensor_to_spec_shape(arr, spec)
        rows.append(arr)
      stacked = np.stack(rows)
      return self._to_torch(stacked, spec)
    rows = []
    expect = _spec_elements(spec)
    for v in values:
      arr = np.asarray(v)
      if arr.size != expect:
        raise ValueError(
            f"Feature {key}: got {arr.size} values, spec {spec.shape} wants "
            f"{expect}")
      rows.append(arr.reshape([d if d is not None else -1
                               for d in spec.shape] or ()))
    return self._to_torch(np.stack(rows), spec)

  def _assemble_sequence(self, key, spec, values) -> torch.Tensor:
    max_len = max(len(steps) for steps in values) if values else 0
    shape = tuple(d for d in spec.shape)
    np_dt = _np_parse_dtype(spec)
    if tsu.is_encoded_image_spec(spec):
      single = tuple(spec.shape[-3:])
      out = np.zeros((len(values), max_len) + single, dtype=spec.np_dtype)
      for b, steps in enumerate(values):
        flat = [s[0] if s else b"" for s in steps]
        out[b, : len(steps)] = _decode_image_batch(flat, spec)
      return torch.from_numpy(out)
    out = np.zeros((len(values), max_len) + shape, dtype=np_dt)
    expect = _spec_elements(spec)
    for b, steps in enumerate(values):
      for t, v in enumerate(steps):
        arr = np.asarray(v)
        if arr.size != expect:
          raise ValueError(
              f"Sequence feature {key} step {t}: {arr.size} values, want "
              f"{expect}")
        out[b, t] = arr.reshape(shape or ())
    return self._to_torch(out, spec)

  @staticmethod
  def _to_torch(arr: np.ndarray, spec) -> torch.Tensor:
    t = torch.from_numpy(np.ascontiguousarray(arr))
    if spec.dtype in (torch.bfloat16, torch.float16) and \
        t.dtype == torch.float32:
      t = t.to(spec.dtype)  # bf16 specs parsed as f32, cast (:326-391)
    return t


def create_parse_example_fn(feature_spec, label_spec=None,
                            decode_images: bool = True):
  """Builds fn({dataset_key: [bytes]}) -> (features, labels) structs."""
  flat_features = tsu.flatten_spec_structure(feature_spec)
  flat_labels = tsu.flatten_spec_structure(label_spec) if label_spec \
      is not None else tsu.TensorSpecStruct()
  tsu.assert_valid_spec_structure(flat_features)
  tsu.assert_valid_spec_structure(flat_labels)

  dataset_keys = sorted({s.dataset_key for s in flat_features.values()} |
                        {s.dataset_key for s in flat_labels.values()})
  parsers = {}
  for dkey in dataset_keys:
    merged = tsu.TensorSpecStruct()
    for prefix, flat in (("features", flat_features), ("labels", flat_labels)):
      for key, spec in flat.items():
        if (spec.dataset_key or "") == dkey:
          merged[prefix + "/" + key] = spec
    parsers[dkey] = ExampleParser(merged, decode_images=decode_images)

  def parse(batched_records) -> Tuple[tsu.TensorSpecStruct,
                                      Optional[tsu.TensorSpecStruct]]:
    if isinstance(batched_records, (list, tuple)):
      batched_records = {"": list(batched_records)}
    parsed = tsu.TensorSpecStruct()
    for dkey, parser in parsers.items():
      if dkey not in batched_records:
        raise ValueError(
            f"No record batch for dataset_key {dkey!r}; got "
            f"{list(batched_records.keys())}")
      for key, value in parser(batched_records[dkey]).items():
        parsed[key] = value

    def collect(prefix, flat_spec):
      out = tsu.TensorSpecStruct()
      for key, spec in flat_spec.items():
        pkey = prefix + "/" + key
        if pkey in parsed:
          out[key] = parsed[pkey]
          lkey = pkey + "_length"
          if spec.is_sequence and lkey in parsed:
            out[key + "_length"] = parsed[lkey]
        elif not spec.is_optional:
          raise ValueError(f"Missing parsed tensor for required {key}")
      return out

    features = collect("features", flat_features)
    labels = collect("labels", flat_labels) if flat_labels else None
    return features, labels

  return parse
