"""Typed tensor contracts: ExtendedTensorSpec and TensorSpecStruct.

The spine of the framework: every layer (data pipeline, preprocessors,
models, export, predictors) communicates through structures of
ExtendedTensorSpec.  Re-designed for PyTorch from the behavior of the
reference's `utils/tensorspec_utils.py` (ExtendedTensorSpec :40,
TensorSpecStruct :302, flatten :1303, pack :1348, validate_and_flatten :1210,
validate_and_pack :1244, feature-dict generation :1558-1629, varlen pad/clip
:1631, dtype rewrites :690-752, assets IO :1685).
"""

from __future__ import annotations

import collections
from typing import Any, Dict, Mapping, Optional, Sequence, Tuple

import numpy as np
import torch

from tensor2robot_amd import ginconfig as gin

# ---------------------------------------------------------------------------
# dtype registry: canonical torch dtype <-> string <-> numpy
# ---------------------------------------------------------------------------

_DTYPE_BY_NAME = {
    "float16": torch.float16,
    "half": torch.float16,
    "bfloat16": torch.bfloat16,
    "float32": torch.float32,
    "float": torch.float32,
    "float64": torch.float64,
    "double": torch.float64,
    "uint8": torch.uint8,
    "uint16": torch.int32,  # torch has no uint16 math type; widen.
    "int8": torch.int8,
    "int16": torch.int16,
    "int32": torch.int32,
    "int64": torch.int64,
    "bool": torch.bool,
    "string": torch.uint8,  # byte-string features surface as uint8 buffers
}

_NAME_BY_DTYPE = {
    torch.float16: "float16",
    torch.bfloat16: "bfloat16",
    torch.float32: "float32",
    torch.float64: "float64",
    torch.uint8: "uint8",
    torch.int8: "int8",
    torch.int16: "int16",
    torch.int32: "int32",
    torch.int64: "int64",
    torch.bool: "bool",
}

_NUMPY_BY_DTYPE = {
    torch.float16: np.float16,
    torch.bfloat16: np.float32,  # numpy has no bf16; host side stays f32
    torch.float32: np.float32,
    torch.float64: np.float64,
    torch.uint8: np.uint8,
    torch.int8: np.int8,
    torch.int16: np.int16,
    torch.int32: np.int32,
    torch.int64: np.int64,
    torch.bool: np.bool_,
}

_TORCH_BY_NUMPY = {
    np.dtype(np.float16): torch.float16,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.int8): torch.int8,
    np.dtype(np.int16): torch.int16,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.bool_): torch.bool,
}


def canonical_dtype(dtype) -> torch.dtype:
  """Accepts torch dtype, numpy dtype/type, or string name."""
  if isinstance(dtype, torch.dtype):
    return dtype
  if isinstance(dtype, str):
    if dtype not in _DTYPE_BY_NAME:
      raise ValueError(f"Unknown dtype name {dtype!r}")
    return _DTYPE_BY_NAME[dtype]
  try:
    np_dtype = np.dtype(dtype)
  except TypeError as e:
    raise ValueError(f"Cannot interpret dtype {dtype!r}") from e
  if np_dtype in _TORCH_BY_NUMPY:
    return _TORCH_BY_NUMPY[np_dtype]
  raise ValueError(f"Unsupported dtype {dtype!r}")


def dtype_name(dtype: torch.dtype) -> str:
  return _NAME_BY_DTYPE[canonical_dtype(dtype)]


def numpy_dtype(dtype) -> np.dtype:
  return np.dtype(_NUMPY_BY_DTYPE[canonical_dtype(dtype)])


# ---------------------------------------------------------------------------
# ExtendedTensorSpec
# ---------------------------------------------------------------------------

_ALLOWED_DATA_FORMATS = (None, "jpeg", "JPEG", "png", "PNG")


class ExtendedTensorSpec:
  """A tensor contract: shape + dtype + framework metadata.

  Mirrors the reference's semantics (`utils/tensorspec_utils.py:40-278`):
    is_optional: tensor may be absent; pack fills None.
    is_sequence: parsed from the sequence (FixedLenSequenceFeature) half of a
      SequenceExample; gets a companion `<name>_length` tensor.
    is_extracted: marks specs derived from actual tensors.
    data_format: 'JPEG'/'PNG' => serialized image string to be decoded.
    dataset_key: multi-dataset routing key.
    varlen_default_value: if set, feature parses as variable-length and is
      padded with this value / clipped to shape[0].
  """

  __slots__ = ("_shape", "_dtype", "_name", "_is_optional", "_is_sequence",
               "_is_extracted", "_data_format", "_dataset_key",
               "_varlen_default_value")

  def __init__(self, shape, dtype, name=None, is_optional=None,
               is_sequence=False, is_extracted=False, data_format=None,
               dataset_key="", varlen_default_value=None):
    if shape is None:
      shape = ()
    if isinstance(shape, (int, np.integer)):
      shape = (int(shape),)
    self._shape = tuple(
        None if d is None else int(d) for d in shape)
    for d in self._shape:
      if d is not None and d < -1:
        raise ValueError(f"Invalid dimension {d} in shape {shape}")
    self._dtype = canonical_dtype(dtype)
    self._name = name
    self._is_optional = bool(is_optional) if is_optional is not None else False
    self._is_sequence = bool(is_sequence)
    self._is_extracted = bool(is_extracted)
    if data_format is not None and data_format.upper() not in ("JPEG", "PNG"):
      raise ValueError(f"Unsupported data_format {data_format!r}")
    self._data_format = data_format
    self._dataset_key = dataset_key or ""
    if varlen_default_value is not None and not self._shape:
      raise ValueError("varlen_default_value requires a non-scalar shape "
                       "(shape[0] is the max length).")
    self._varlen_default_value = varlen_default_value

  # -- properties ----------------------------------------------------------
  @property
  def shape(self) -> Tuple[Optional[int], ...]:
    return self._shape

  @property
  def dtype(self) -> torch.dtype:
    return self._dtype

  @property
  def name(self) -> Optional[str]:
    return self._name

  @property
  def is_optional(self) -> bool:
    return self._is_optional

  @property
  def is_sequence(self) -> bool:
    return self._is_sequence

  @property
  def is_extracted(self) -> bool:
    return self._is_extracted

  @property
  def data_format(self) -> Optional[str]:
    return self._data_format

  @property
  def dataset_key(self) -> str:
    return self._dataset_key

  @property
  def varlen_default_value(self):
    return self._varlen_default_value

  @property
  def np_dtype(self) -> np.dtype:
    return numpy_dtype(self._dtype)

  # -- constructors --------------------------------------------------------
  @classmethod
  def from_spec(cls, spec, shape=None, dtype=None, name=None,
                is_optional=None, is_sequence=None, is_extracted=None,
                data_format=None, dataset_key=None, batch_size=None,
                varlen_default_value=None):
    """Copies a spec, overriding fields; batch_size prepends a batch dim.

    batch_size semantics (reference :144-153): a positive batch_size prepends
    that dim; batch_size=-1 prepends a dynamic (None) dim; batch_size=None
    leaves the shape unchanged.
    """
    new_shape = tuple(spec.shape) if shape is None else tuple(shape)
    if batch_size is not None:
      if batch_size == -1:
        new_shape = (None,) + new_shape
      else:
        new_shape = (int(batch_size),) + new_shape
    get = lambda override, current: current if override is None else override
    return cls(
        shape=new_shape,
        dtype=get(dtype, spec.dtype),
        name=get(name, getattr(spec, "name", None)),
        is_optional=get(is_optional, getattr(spec, "is_optional", False)),
        is_sequence=get(is_sequence, getattr(spec, "is_sequence", False)),
        is_extracted=get(is_extracted, getattr(spec, "is_extracted", False)),
        data_format=get(data_format, getattr(spec, "data_format", None)),
        dataset_key=get(dataset_key, getattr(spec, "dataset_key", "")),
        varlen_default_value=get(
            varlen_default_value, getattr(spec, "varlen_default_value", None)),
    )

  @classmethod
  def from_tensor(cls, tensor, name=None):
    if isinstance(tensor, np.ndarray):
      return cls(shape=tensor.shape, dtype=canonical_dtype(tensor.dtype),
                 name=name, is_extracted=True)
    if isinstance(tensor, torch.Tensor):
      return cls(shape=tuple(tensor.shape), dtype=tensor.dtype, name=name,
                 is_extracted=True)
    raise ValueError(f"Cannot build spec from {type(tensor)}")

  @classmethod
  def to_spec(cls, instance):
    """Normalizes a tensor or spec to an ExtendedTensorSpec."""
    if isinstance(instance, ExtendedTensorSpec):
      return instance
    if isinstance(instance, (np.ndarray, torch.Tensor)):
      return cls.from_tensor(instance)
    raise ValueError(f"Cannot convert {type(instance)} to spec")

  # -- serialization (proto-text round trip) -------------------------------
  def to_proto_dict(self) -> Dict[str, Any]:
    d = {"shape": list(-1 if s is None else s for s in self._shape),
         "dtype": dtype_name(self._dtype)}
    if self._name:
      d["name"] = self._name
    if self._is_optional:
      d["is_optional"] = True
    if self._is_sequence:
      d["is_sequence"] = True
    if self._is_extracted:
      d["is_extracted"] = True
    if self._data_format:
      d["data_format"] = self._data_format
    if self._dataset_key:
      d["dataset_key"] = self._dataset_key
    if self._varlen_default_value is not None:
      d["varlen_default_value"] = float(self._varlen_default_value)
    return d

  @classmethod
  def from_proto_dict(cls, d: Mapping[str, Any]) -> "ExtendedTensorSpec":
    raw_shape = d.get("shape", [])
    if isinstance(raw_shape, (int, np.integer)):
      raw_shape = [raw_shape]  # pbtxt collapses single repeated field
    shape = tuple(None if s == -1 else int(s) for s in raw_shape)
    return cls(shape=shape, dtype=d.get("dtype", "float32"),
               name=d.get("name"), is_optional=d.get("is_optional", False),
               is_sequence=d.get("is_sequence", False),
               is_extracted=d.get("is_extracted", False),
               data_format=d.get("data_format"),
               dataset_key=d.get("dataset_key", ""),
               varlen_default_value=d.get("varlen_default_value"))

  # -- dunder --------------------------------------------------------------
  def __eq__(self, other):
    """Equality is shape+dtype only (reference :261)."""
    if not isinstance(other, ExtendedTensorSpec):
      return NotImplemented
    return self._shape == other._shape and self._dtype == other._dtype

  def __ne__(self, other):
    eq = self.__eq__(other)
    return NotImplemented if eq is NotImplemented else not eq

  def __hash__(self):
    return hash((self._shape, self._dtype))

  def __repr__(self):
    extras = []
    if self._name:
      extras.append(f"name={self._name!r}")
    if self._is_optional:
      extras.append("is_optional=True")
    if self._is_sequence:
      extras.append("is_sequence=True")
    if self._data_format:
      extras.append(f"data_format={self._data_format!r}")
    if self._dataset_key:
      extras.append(f"dataset_key={self._dataset_key!r}")
    if self._varlen_default_value is not None:
      extras.append(f"varlen_default_value={self._varlen_default_value!r}")
    extra = (", " + ", ".join(extras)) if extras else ""
    return (f"ExtendedTensorSpec(shape={self._shape}, "
            f"dtype={dtype_name(self._dtype)}{extra})")

  def is_compatible_with(self, value) -> bool:
    """Shape/dtype compatibility with a tensor/ndarray/spec (None matches)."""
    if isinstance(value, ExtendedTensorSpec):
      other_shape, other_dtype = value.shape, value.dtype
    elif isinstance(value, (np.ndarray, torch.Tensor)):
      other_shape = tuple(value.shape)
      other_dtype = canonical_dtype(
          value.dtype if isinstance(value, torch.Tensor) else value.dtype)
    else:
      return False
    if other_dtype != self._dtype:
      return False
    if len(other_shape) != len(self._shape):
      return False
    for mine, theirs in zip(self._shape, other_shape):
      if mine is not None and theirs is not None and mine != theirs:
        return False
    return True


TensorSpec = ExtendedTensorSpec  # alias


# ---------------------------------------------------------------------------
# TensorSpecStruct: flat + hierarchical ordered mapping with live views
# ---------------------------------------------------------------------------


class TensorSpecStruct(collections.OrderedDict):
  """An ordered dict that is simultaneously flat and hierarchical.

  Keys are '/'-joined paths ('train/images').  Attribute access returns live
  sub-views that share the parent storage: mutating `s.train.images` mutates
  `s['train/images']` (reference `utils/tensorspec_utils.py:302-683`).
  """

  def __init__(self, *args, **kwargs):
    object.__setattr__(self, "_root", None)
    object.__setattr__(self, "_prefix", "")
    super().__init__()
    init = collections.OrderedDict(*args, **kwargs)
    for k, v in init.items():
      self[k] = v

  # -- view plumbing -------------------------------------------------------
  @classmethod
  def _make_view(cls, root: "TensorSpecStruct", prefix: str):
    view = cls.__new__(cls)
    collections.OrderedDict.__init__(view)
    object.__setattr__(view, "_root", root)
    object.__setattr__(view, "_prefix", prefix)
    return view

  def _storage(self):
    return self._root if self._root is not None else self

  def _abs_key(self, key):
    if self._prefix:
      return self._prefix + "/" + key
    return key

  @staticmethod
  def _check_key(key):
    if not isinstance(key, str) or not key:
      raise ValueError(f"TensorSpecStruct keys must be non-empty strings, "
                       f"got {key!r}")
    for part in key.split("/"):
      if not part:
        raise ValueError(f"Empty path component in key {key!r}")

  # -- mapping interface ---------------------------------------------------
  def __getitem__(self, key):
    self._check_key(key)
    store = self._storage()
    abs_key = self._abs_key(key)
    if collections.OrderedDict.__contains__(store, abs_key):
      return collections.OrderedDict.__getitem__(store, abs_key)
    # Prefix lookup: build a sub-view on demand (reference :437-486).
    prefix = abs_key + "/"
    if any(k.startswith(prefix)
           for k in collections.OrderedDict.keys(store)):
      return TensorSpecStruct._make_view(store, abs_key)
    raise KeyError(key)

  def __setitem__(self, key, value):
    self._check_key(key)
    store = self._storage()
    abs_key = self._abs_key(key)
    if isinstance(value, (dict, TensorSpecStruct)):
      # Setting a sub-structure: flatten it under this prefix.
      sub = flatten_spec_structure(value)
      for k, v in sub.items():
        collections.OrderedDict.__setitem__(store, abs_key + "/" + k, v)
      return
    collections.OrderedDict.__setitem__(store, abs_key, value)

  def __delitem__(self, key):
    store = self._storage()
    abs_key = self._abs_key(key)
    if collections.OrderedDict.__contains__(store, abs_key):
      collections.OrderedDict.__delitem__(store, abs_key)
      return
    prefix = abs_key + "/"
    sub = [k for k in list(collections.OrderedDict.keys(store))
           if k.startswith(prefix)]
    if not sub:
      raise KeyError(key)
    for k in sub:
      collections.OrderedDict.__delitem__(store, k)

  def __contains__(self, key):
    try:
      self[key]
      return True
    except (KeyError, ValueError):
      return False

  def _rel_keys(self):
    store = self._storage()
    if not self._prefix:
      return list(collections.OrderedDict.keys(store))
    prefix = self._prefix + "/"
    return [k[len(prefix):]
            for k in collections.OrderedDict.keys(store)
            if k.startswith(prefix)]

  def keys(self):
    return self._rel_keys()

  def __iter__(self):
    return iter(self._rel_keys())

  def __len__(self):
    return len(self._rel_keys())

  def values(self):
    return [self[k] for k in self._rel_keys()]

  def items(self):
    return [(k, self[k]) for k in self._rel_keys()]

  def get(self, key, default=None):
    try:
      return self[key]
    except KeyError:
      return default

  def update(self, other=(), **kwargs):
    if hasattr(other, "items"):
      other = other.items()
    for k, v in other:
      self[k] = v
    for k, v in kwargs.items():
      self[k] = v

  def pop(self, key, *default):
    try:
      value = self[key]
    except KeyError:
      if default:
        return default[0]
      raise
    del self[key]
    return value

  # -- attribute access ----------------------------------------------------
  def __getattr__(self, name):
    if name.startswith("_") or name in ("keys", "items", "values"):
      raise AttributeError(name)
    try:
      return self[name]
    except KeyError:
      raise AttributeError(
          f"TensorSpecStruct has no key or sub-structure {name!r}; "
          f"keys: {self._rel_keys()}") from None

  def __setattr__(self, name, value):
    if name.startswith("_"):
      object.__setattr__(self, name, value)
    else:
      self[name] = value

  def __delattr__(self, name):
    if name.startswith("_"):
      object.__delattr__(self, name)
    else:
      del self[name]

  # -- conversions ---------------------------------------------------------
  def to_dict(self) -> Dict[str, Any]:
    return collections.OrderedDict(self.items())

  def to_nested_dict(self) -> Dict[str, Any]:
    out = collections.OrderedDict()
    for key, value in self.items():
      parts = key.split("/")
      node = out
      for p in parts[:-1]:
        node = node.setdefault(p, collections.OrderedDict())
      node[parts[-1]] = value
    return out

  def __reduce__(self):
    return (_rebuild_struct, (self.to_dict(),))

  def __repr__(self):
    inner = ", ".join(f"{k!r}: {v!r}" for k, v in self.items())
    return f"TensorSpecStruct({{{inner}}})"

  def __eq__(self, other):
    if isinstance(other, (dict, TensorSpecStruct)):
      other_items = list(flatten_spec_structure(other).items()) if not \
          isinstance(other, TensorSpecStruct) else list(other.items())
      return list(self.items()) == other_items
    return NotImplemented

  def __ne__(self, other):
    eq = self.__eq__(other)
    return NotImplemented if eq is NotImplemented else not eq

  def copy(self):
    return TensorSpecStruct(self.items())


def _rebuild_struct(d):
  return TensorSpecStruct(d)


# ---------------------------------------------------------------------------
# flatten / pack / validate
# ---------------------------------------------------------------------------


def _is_leaf(value):
  return not isinstance(value, (dict, TensorSpecStruct)) or isinstance(
      value, np.ndarray)


def flatten_spec_structure(spec_structure) -> TensorSpecStruct:
  """Flattens nested dicts/namedtuples/TensorSpecStructs into path keys.

  None values for optional specs are dropped (reference :1303-1346 filters
  None optionals).
  """
  flat = TensorSpecStruct()

  def visit(prefix, value):
    if isinstance(value, TensorSpecStruct):
      for k, v in value.items():
        visit(prefix + (k,), v)
      return
    if isinstance(value, Mapping):
      for k, v in value.items():
        visit(prefix + (str(k),), v)
      return
    if hasattr(value, "_asdict"):  # namedtuple
      for k, v in value._asdict().items():
        visit(prefix + (k,), v)
      return
    if value is None:
      return  # dropped optional
    key = "/".join(prefix)
    collections.OrderedDict.__setitem__(flat, key, value)

  if _is_leaf(spec_structure) and not isinstance(spec_structure, Mapping) \
      and not hasattr(spec_structure, "_asdict"):
    raise ValueError(
        f"flatten_spec_structure expects a structure, got "
        f"{type(spec_structure)}")
  visit((), spec_structure)
  return flat


def pack_flat_sequence_to_spec_structure(spec_structure,
                                         flat_tensors) -> TensorSpecStruct:
  """Packs flat tensors into the layout of spec_structure.

  Missing optional entries are skipped; missing required entries raise
  (reference :1348-1432).
  """
  specs = flatten_spec_structure(spec_structure)
  flat = flat_tensors if isinstance(flat_tensors, TensorSpecStruct) else \
      flatten_spec_structure(flat_tensors)
  packed = TensorSpecStruct()
  for key, spec in specs.items():
    if key in flat:
      packed[key] = flat[key]
    else:
      if isinstance(spec, ExtendedTensorSpec) and spec.is_optional:
        continue
      raise ValueError(
          f"Required spec '{key}' has no matching tensor; available: "
          f"{list(flat.keys())}")
  return packed


def assert_required(spec_structure):
  """Raises if any non-optional spec is None-valued."""
  for key, spec in flatten_spec_structure(spec_structure).items():
    if spec is None:
      raise ValueError(f"Required spec {key} is None")


def maybe_ignore_batch(shape, ignore_batch):
  if ignore_batch:
    return tuple(shape)[1:]
  return tuple(shape)


def _value_shape_dtype(value):
  if isinstance(value, ExtendedTensorSpec):
    return value.shape, value.dtype
  if isinstance(value, torch.Tensor):
    return tuple(value.shape), value.dtype
  if isinstance(value, np.ndarray):
    return tuple(value.shape), canonical_dtype(value.dtype)
  raise ValueError(f"Cannot check type {type(value)}")


def assert_equal_spec_or_tensor(expected_spec, actual, ignore_batch=False):
  e_shape, e_dtype = _value_shape_dtype(expected_spec)
  a_shape, a_dtype = _value_shape_dtype(actual)
  if ignore_batch:
    # Specs are usually batchless while tensors carry a leading batch dim:
    # strip the batch from whichever side has it.
    if len(a_shape) == len(e_shape) + 1:
      a_shape = a_shape[1:]
    elif len(e_shape) == len(a_shape) + 1:
      e_shape = e_shape[1:]
    elif len(e_shape) == len(a_shape) and e_shape:
      e_shape, a_shape = e_shape[1:], a_shape[1:]
  if e_dtype != a_dtype:
    raise ValueError(
        f"dtype mismatch: expected {e_dtype}, got {a_dtype}")
  if len(e_shape) != len(a_shape):
    raise ValueError(
        f"rank mismatch: expected {e_shape}, got {a_shape}")
  for e, a in zip(e_shape, a_shape):
    if e is not None and a is not None and e != a:
      raise ValueError(f"shape mismatch: expected {e_shape}, got {a_shape}")


def assert_equal(expected_spec_structure, actual_structure,
                 ignore_batch=False):
  """Structure-wise spec equality (reference :1142-1167)."""
  expected = flatten_spec_structure(expected_spec_structure)
  actual = flatten_spec_structure(actual_structure)
  for key, spec in expected.items():
    is_opt = isinstance(spec, ExtendedTensorSpec) and spec.is_optional
    if key not in actual:
      if is_opt:
        continue
      raise ValueError(
          f"Missing required entry {key!r}; actual keys: "
          f"{list(actual.keys())}")
    assert_equal_spec_or_tensor(spec, actual[key], ignore_batch=ignore_batch)


def validate_and_flatten(expected_spec_structure, actual_tensors_or_spec,
                         ignore_batch=False) -> TensorSpecStruct:
  """Asserts actual matches expected, returns the flat actual structure."""
  assert_equal(expected_spec_structure, actual_tensors_or_spec,
               ignore_batch=ignore_batch)
  return flatten_spec_structure(actual_tensors_or_spec)


def validate_and_pack(expected_spec_structure, actual_tensors_or_spec,
                      ignore_batch=False) -> TensorSpecStruct:
  """Asserts actual matches expected, returns actual packed to expected."""
  assert_equal(expected_spec_structure, actual_tensors_or_spec,
               ignore_batch=ignore_batch)
  return pack_flat_sequence_to_spec_structure(expected_spec_structure,
                                              actual_tensors_or_spec)


def assert_valid_spec_structure(spec_structure):
  """Enforces the unique-or-identical `name` rule (reference :1503-1515).

  Two specs may share a `name` only if their (shape, dtype) are identical —
  they then map to the same serialized feature.
  """
  seen: Dict[str, ExtendedTensorSpec] = {}
  for key, spec in flatten_spec_structure(spec_structure).items():
    if not isinstance(spec, ExtendedTensorSpec):
      raise ValueError(
          f"Entry {key} is not an ExtendedTensorSpec: {type(spec)}")
    name = spec.name
    if name is None:
      continue
    dedup_key = spec.dataset_key + ":" + name
    if dedup_key in seen:
      prev = seen[dedup_key]
      if prev.shape != spec.shape or prev.dtype != spec.dtype:
        raise ValueError(
            f"Specs sharing name {name!r} disagree: {prev} vs {spec}")
    else:
      seen[dedup_key] = spec


def filter_spec_structure_by_dataset(spec_structure,
                                     dataset_key) -> TensorSpecStruct:
  """Keeps only specs routed to dataset_key (reference :1291)."""
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    if isinstance(spec, ExtendedTensorSpec) and \
        (spec.dataset_key or "") == (dataset_key or ""):
      out[key] = spec
  return out


def filter_required_flat_tensor_spec(flat_spec) -> TensorSpecStruct:
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(flat_spec).items():
    if isinstance(spec, ExtendedTensorSpec) and spec.is_optional:
      continue
    out[key] = spec
  return out


def copy_tensorspec(spec_structure, batch_size=None,
                    prefix="") -> TensorSpecStruct:
  """Deep-copies a spec structure, optionally re-batching and renaming."""
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    name = spec.name
    if prefix and name:
      name = prefix + "/" + name
    out[key] = ExtendedTensorSpec.from_spec(spec, batch_size=batch_size,
                                            name=name)
  return out


# ---------------------------------------------------------------------------
# Random / constant tensor factories (test + export backbone)
# ---------------------------------------------------------------------------


def _concrete_shape(spec, batch_size=None, sequence_length=None):
  shape = list(spec.shape)
  if batch_size is not None:
    # Reference :817-925: the batch dim is always PREPENDED; a leading
    # None (e.g. the meta-learning samples dim from batch_size=-1 specs)
    # stays as an inner dim and concretizes to 1 below.
    shape = [batch_size] + shape
  if spec.is_sequence:
    # A sequence spec's data carries an episode/time dim after batch.
    seq = sequence_length if sequence_length is not None else 1
    shape = [shape[0], seq] + shape[1:]
  shape = [1 if d is None else d for d in shape]
  return tuple(int(d) for d in shape)


def make_random_numpy(spec_structure, batch_size=None, sequence_length=None,
                      seed=None):
  """Spec-conformant random numpy feeds (reference :886-921)."""
  rng = np.random.RandomState(seed)
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    shape = _concrete_shape(spec, batch_size, sequence_length)
    npdt = spec.np_dtype
    if np.issubdtype(npdt, np.floating):
      data = rng.uniform(0.0, 1.0, size=shape).astype(npdt)
    elif npdt == np.bool_:
      data = rng.uniform(size=shape) > 0.5
    else:
      info = np.iinfo(npdt)
      high = min(info.max, 255)
      data = rng.randint(0, high + 1, size=shape).astype(npdt)
    out[key] = data
  return out


def make_constant_numpy(spec_structure, constant_value=0.0, batch_size=None,
                        sequence_length=None):
  """Spec-conformant constant numpy feeds (reference :847-884)."""
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    shape = _concrete_shape(spec, batch_size, sequence_length)
    out[key] = np.full(shape, constant_value, dtype=spec.np_dtype)
  return out


def make_random_tensors(spec_structure, batch_size=None, sequence_length=None,
                        device="cpu", seed=None):
  """Spec-conformant random torch tensors (placeholder analog :783-845)."""
  gen = torch.Generator(device="cpu")
  if seed is not None:
    gen.manual_seed(seed)
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    shape = _concrete_shape(spec, batch_size, sequence_length)
    if spec.dtype.is_floating_point:
      t = torch.rand(shape, generator=gen, dtype=torch.float32).to(spec.dtype)
    elif spec.dtype == torch.bool:
      t = torch.rand(shape, generator=gen) > 0.5
    else:
      t = torch.randint(0, 255, shape, generator=gen).to(spec.dtype)
    out[key] = t.to(device)
  return out


def map_feed_dict(spec_structure, numpy_struct, ignore_batch=False):
  """Validated {flat_key: ndarray} mapping for predictors (reference :923)."""
  flat_np = flatten_spec_structure(numpy_struct)
  assert_equal(spec_structure, flat_np, ignore_batch=ignore_batch)
  return {k: v for k, v in flat_np.items()}


# ---------------------------------------------------------------------------
# tf.Example schema generation (consumed by the native parser)
# ---------------------------------------------------------------------------


class FeatureKind:
  FIXED_LEN = "fixed_len"
  FIXED_LEN_SEQUENCE = "fixed_len_sequence"
  VARLEN = "varlen"


class FeatureSchema(
    collections.namedtuple(
        "FeatureSchema",
        ["kind", "shape", "dtype", "is_image", "varlen_default_value"])):
  """One serialized-feature descriptor derived from a spec."""


def is_encoded_image_spec(spec: ExtendedTensorSpec) -> bool:
  return spec.data_format is not None and \
      spec.data_format.upper() in ("JPEG", "PNG")


def tensorspec_to_feature_schema(spec: ExtendedTensorSpec) -> FeatureSchema:
  """Spec -> parse schema entry (reference `_get_feature` :1571-1594)."""
  if is_encoded_image_spec(spec):
    # Encoded images are stored as byte strings.
    kind = (FeatureKind.FIXED_LEN_SEQUENCE if spec.is_sequence
            else FeatureKind.FIXED_LEN)
    return FeatureSchema(kind=kind, shape=(), dtype="string", is_image=True,
                         varlen_default_value=None)
  if spec.varlen_default_value is not None:
    return FeatureSchema(kind=FeatureKind.VARLEN, shape=tuple(spec.shape),
                         dtype=dtype_name(spec.dtype), is_image=False,
                         varlen_default_value=spec.varlen_default_value)
  kind = (FeatureKind.FIXED_LEN_SEQUENCE if spec.is_sequence
          else FeatureKind.FIXED_LEN)
  return FeatureSchema(kind=kind, shape=tuple(spec.shape),
                       dtype=dtype_name(spec.dtype), is_image=False,
                       varlen_default_value=None)


def tensorspec_to_feature_dict(spec_structure, decode_images=True):
  """Flat {serialized-name: FeatureSchema} plus name->key mapping.

  Returns (schema_dict, key_by_name) where key_by_name maps each serialized
  feature name back to the flat spec keys that consume it (reference
  :1596-1629).
  """
  assert_valid_spec_structure(spec_structure)
  schema = {}
  keys_by_name = collections.defaultdict(list)
  for key, spec in flatten_spec_structure(spec_structure).items():
    name = spec.name or key
    schema[name] = tensorspec_to_feature_schema(spec)
    keys_by_name[name].append(key)
  return schema, dict(keys_by_name)


def pad_or_clip_tensor_to_spec_shape(tensor, spec):
  """Pads (with varlen_default_value) or clips dim0 to spec.shape[0].

  Reference :1631-1682.  Works on torch tensors and numpy arrays.
  """
  target = spec.shape[0]
  if target is None:
    return tensor
  length = tensor.shape[0]
  if length == target:
    return tensor
  if length > target:
    return tensor[:target]
  pad_len = target - length
  fill = spec.varlen_default_value
  if fill is None:
    fill = 0
  if isinstance(tensor, np.ndarray):
    pad = np.full((pad_len,) + tensor.shape[1:], fill, dtype=tensor.dtype)
    return np.concatenate([tensor, pad], axis=0)
  pad = torch.full((pad_len,) + tuple(tensor.shape[1:]), fill,
                   dtype=tensor.dtype, device=tensor.device)
  return torch.cat([tensor, pad], dim=0)


# ---------------------------------------------------------------------------
# dtype rewrites (bf16 device discipline)
# ---------------------------------------------------------------------------


def replace_dtype(spec_structure, from_dtype, to_dtype) -> TensorSpecStruct:
  """Spec-level dtype rewrite (reference :690-711)."""
  from_dtype = canonical_dtype(from_dtype)
  to_dtype = canonical_dtype(to_dtype)
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    if spec.dtype == from_dtype:
      out[key] = ExtendedTensorSpec.from_spec(spec, dtype=to_dtype)
    else:
      out[key] = spec
  return out


def cast_float32_to_bfloat16(tensor_struct, output_spec_structure):
  """Casts f32 tensors whose spec declares bf16 (reference :713-736)."""
  specs = flatten_spec_structure(output_spec_structure)
  tensors = flatten_spec_structure(tensor_struct)
  out = TensorSpecStruct()
  for key, t in tensors.items():
    spec = specs.get(key)
    if spec is not None and spec.dtype == torch.bfloat16 and \
        isinstance(t, torch.Tensor) and t.dtype == torch.float32:
      out[key] = t.to(torch.bfloat16)
    else:
      out[key] = t
  return out


def cast_bfloat16_to_float32(tensor_struct):
  """Casts every bf16 tensor to f32 (reference :738-752)."""
  tensors = flatten_spec_structure(tensor_struct)
  out = TensorSpecStruct()
  for key, t in tensors.items():
    if isinstance(t, torch.Tensor) and t.dtype == torch.bfloat16:
      out[key] = t.to(torch.float32)
    else:
      out[key] = t
  return out


# ---------------------------------------------------------------------------
# T2R assets IO (pbtxt; export <-> predictor handshake)
# ---------------------------------------------------------------------------

T2R_ASSETS_FILENAME = "t2r_assets.pbtxt"
EXTRA_ASSETS_DIRECTORY = "assets.extra"


class T2RAssets:
  """Feature/label spec + global_step bundle shipped with every export."""

  def __init__(self, feature_spec=None, label_spec=None, global_step=0):
    self.feature_spec = feature_spec
    self.label_spec = label_spec
    self.global_step = int(global_step)

  def to_pbtxt(self) -> str:
    from tensor2robot_amd.proto import pbtxt
    d = {"global_step": self.global_step}
    if self.feature_spec is not None:
      d["feature_spec"] = _spec_struct_to_proto_dict(self.feature_spec)
    if self.label_spec is not None:
      d["label_spec"] = _spec_struct_to_proto_dict(self.label_spec)
    return pbtxt.dumps(d)

  @classmethod
  def from_pbtxt(cls, text: str) -> "T2RAssets":
    from tensor2robot_amd.proto import pbtxt
    d = pbtxt.loads(text)
    out = cls(global_step=d.get("global_step", 0))
    if "feature_spec" in d:
      out.feature_spec = _proto_dict_to_spec_struct(d["feature_spec"])
    if "label_spec" in d:
      out.label_spec = _proto_dict_to_spec_struct(d["label_spec"])
    return out


def _spec_struct_to_proto_dict(spec_structure):
  entries = []
  for key, spec in flatten_spec_structure(spec_structure).items():
    e = {"key": key}
    e["spec"] = spec.to_proto_dict()
    entries.append(e)
  return {"key_value": entries}


def _proto_dict_to_spec_struct(d):
  out = TensorSpecStruct()
  entries = d.get("key_value", [])
  if isinstance(entries, dict):
    entries = [entries]
  for e in entries:
    out[e["key"]] = ExtendedTensorSpec.from_proto_dict(e.get("spec", {}))
  return out


def write_t2r_assets_to_file(t2r_assets: T2RAssets, path: str):
  """Atomic write of the assets pbtxt (reference :1685-1690)."""
  import os
  tmp = path + ".tmp"
  with open(tmp, "w") as f:
    f.write(t2r_assets.to_pbtxt())
  os.replace(tmp, path)


def load_t2r_assets_from_file(path: str) -> T2RAssets:
  with open(path) as f:
    return T2RAssets.from_pbtxt(f.read())


# ---------------------------------------------------------------------------
# Legacy pickle assets IO + converter (reference :1703-1732 and
# utils/convert_pkl_assets_to_proto_assets.py:35-57)
# ---------------------------------------------------------------------------

T2R_ASSETS_FILENAME = "t2r_assets.pbtxt"
INPUT_SPEC_PKL_FILENAME = "input_specifications.pkl"


def write_input_spec_to_pkl_file(path: str, feature_spec, label_spec):
  """Legacy pickle of (feature, label) specs (reference :1703-1714)."""
  import pickle
  payload = {
      "feature_spec": [
          (k, tuple(v.shape), str(v.dtype), v.name, v.is_optional,
           v.is_sequence, v.data_format, v.dataset_key)
          for k, v in flatten_spec_structure(feature_spec).items()],
      "label_spec": [
          (k, tuple(v.shape), str(v.dtype), v.name, v.is_optional,
           v.is_sequence, v.data_format, v.dataset_key)
          for k, v in flatten_spec_structure(label_spec).items()],
  }
  tmp = path + ".tmp"
  with open(tmp, "wb") as f:
    pickle.dump(payload, f)
  import os
  os.replace(tmp, path)


def load_input_spec_from_pkl_file(path: str):
  """Loads the legacy pickle into (feature, label) spec structs."""
  import pickle
  with open(path, "rb") as f:
    payload = pickle.load(f)

  def unpack(entries):
    out = TensorSpecStruct()
    for (k, shape, dtype, name, is_optional, is_sequence, data_format,
         dataset_key) in entries:
      out[k] = ExtendedTensorSpec(
          shape, canonical_dtype(dtype.replace("torch.", "")), name=name,
          is_optional=is_optional, is_sequence=is_sequence,
          data_format=data_format, dataset_key=dataset_key)
    return out

  return unpack(payload["feature_spec"]), unpack(payload["label_spec"])


def convert_pkl_assets_to_proto_assets(pkl_path: str, assets_path: str,
                                       global_step: int = 0):
  """Pickle -> pbtxt assets converter (reference converter binary)."""
  feature_spec, label_spec = load_input_spec_from_pkl_file(pkl_path)
  assets = T2RAssets(feature_spec=feature_spec, label_spec=label_spec,
                     global_step=global_step)
  write_t2r_assets_to_file(assets, assets_path)
  return assets


# ---------------------------------------------------------------------------
# Reference-named API parity (utils/tensorspec_utils.py): thin entry
# points a reference user would look for, mapped onto the torch-native
# implementations above.
# ---------------------------------------------------------------------------


def convert_to_tensorspecstruct(inputs) -> TensorSpecStruct:
  """Any spec/tensor hierarchy -> flat TensorSpecStruct (reference :686)."""
  if isinstance(inputs, TensorSpecStruct):
    return inputs
  return flatten_spec_structure(inputs)


def make_placeholders(spec_structure, batch_size=None,
                      sequence_length=None) -> TensorSpecStruct:
  """Torch stand-in for TF placeholders (reference :783-813).

  Returns zero tensors shaped per spec: torch has no symbolic batch
  dim, so batch_size None concretizes to 1; batch_size <= 0 omits the
  batch dim; > 0 is fixed — the same three cases the reference's
  placeholder shapes distinguish.
  """
  out = TensorSpecStruct()
  for key, spec in flatten_spec_structure(spec_structure).items():
    if batch_size is not None and batch_size <= 0:
      shape = _concrete_shape(spec, None, sequence_length)
    else:
      shape = _concrete_shape(spec, batch_size or 1, sequence_length)
    out[key] = torch.zeros(shape, dtype=spec.dtype)
  return out


def tensorspec_from_tensors(tensors) -> TensorSpecStruct:
  """Tensor structure -> spec structure with unique names (ref :1043)."""
  out = TensorSpecStruct()
  for i, (key, t) in enumerate(
      flatten_spec_structure(tensors).items()):
    out[key] = ExtendedTensorSpec.from_tensor(t, name=f"{key}/{i}")
  return out


def add_sequence_length_specs(spec_structure) -> TensorSpecStruct:
  """Augments with key + '_length' int64 specs for every sequence spec
  (reference :1280-1289)."""
  flat = flatten_spec_structure(spec_structure)
  out = TensorSpecStruct()
  for key, value in flat.items():
    out[key] = value
    if value.is_sequence:
      out[key + "_length"] = ExtendedTensorSpec(
          shape=(), dtype=torch.int64, name=(value.name or key) +
          "_length")
  return out


def is_flat_spec_or_tensors_structure(spec_or_tensors) -> bool:
  """True iff the structure is already flat {key: leaf} (ref :1430)."""
  if not isinstance(spec_or_tensors, dict):
    return False
  for key, value in spec_or_tensors.items():
    if isinstance(value, dict):
      return False
    if isinstance(key, str) and "/" not in key and isinstance(
        value, (list, tuple)):
      return False
  return True


def map_predict_fn_dict(spec_structure, spec_numpy, feed_dict=None,
                        ignore_batch=False):
  """Builds/extends a validated feed mapping, refusing overwrites
  (reference :968-1010)."""
  flat_np = flatten_spec_structure(spec_numpy)
  flat_spec = flatten_spec_structure(spec_structure)
  assert_required(filter_required_flat_tensor_spec(flat_spec))
  if feed_dict is None:
    feed_dict = {}
  for key, value in flat_np.items():
    if key not in flat_spec:
      continue
    if key in feed_dict:
      raise ValueError(
          f"We would overwrite existing placeholder mapping {key}.")
    spec = flat_spec[key]
    # Specs here carry NO batch (TFModel semantics, reference
    # ignore_batch docstring) and sequences add a time dim: verify the
    # trailing dims against the spec shape.
    sshape = tuple(1 if d is None else d for d in spec.shape)
    vshape = tuple(np.asarray(value).shape)
    if sshape and (len(vshape) < len(sshape) or
                   vshape[len(vshape) - len(sshape):] != sshape):
      raise ValueError(
          f"{key}: shape {vshape} does not end with spec {sshape}")
    feed_dict[key] = value
  return feed_dict


def map_feed_dict_unsafe(feature_placeholders_spec, np_inputs_spec):
  """Unchecked {key: array} mapping (reference :1012-1041); prefer
  map_feed_dict."""
  import logging as _logging
  _logging.getLogger(__name__).warning(
      "map_feed_dict_unsafe is deprecated. Please update to "
      "map_feed_dict.")
  flat_spec = flatten_spec_structure(feature_placeholders_spec)
  flat_np = flatten_spec_structure(np_inputs_spec)
  return {key: flat_np[key] for key in flat_spec if key in flat_np}


# Legacy pickle/pbtxt IO under the reference's exact names
# (reference :1691-1732); the native implementations live above.
def load_t2r_assets_to_file(filename: str) -> "T2RAssets":
  return load_t2r_assets_from_file(filename)


def write_input_spec_to_file(in_feature_spec, in_label_spec,
                             filename: str):
  write_input_spec_to_pkl_file(filename, in_feature_spec, in_label_spec)


def load_input_spec_from_file(filename: str):
  import os as _os
  if not _os.path.exists(filename):
    raise ValueError(f"The file {filename} does not exist.")
  return load_input_spec_from_pkl_file(filename)


def write_global_step_to_file(global_step: int, filename: str):
  import pickle as _pickle
  with open(filename, "wb") as f:
    _pickle.dump({"global_step": int(global_step)}, f)


def load_global_step_from_file(filename: str) -> int:
  import os as _os
  import pickle as _pickle
  if not _os.path.exists(filename):
    raise ValueError(f"The file {filename} does not exist.")
  with open(filename, "rb") as f:
    return _pickle.load(f)["global_step"]
