"""Property-based roundtrip tests of the native tf.Example wire codec.

The hand-written varint/length-delimited encoder (data/example.py,
reference tf.train.Example wire format) must decode every value it can
encode — hypothesis sweeps dtypes, shapes, extreme values and unicode
names far past the hand-picked cases in test_data_pipeline.
"""

import numpy as np
from hypothesis import given, settings, strategies as st

from tensor2robot_amd.data import example as codec


_names = st.text(
    alphabet=st.characters(blacklist_categories=("Cs",), max_codepoint=0x2FF),
    min_size=1, max_size=12)

_float_arrays = st.lists(
    st.floats(width=32, allow_nan=False, allow_infinity=False),
    min_size=0, max_size=8).map(lambda v: np.asarray(v, np.float32))

_int_arrays = st.lists(
    st.integers(min_value=-2**62, max_value=2**62 - 1),
    min_size=0, max_size=8).map(lambda v: np.asarray(v, np.int64))

_bytes_lists = st.lists(st.binary(min_size=0, max_size=20),
                        min_size=0, max_size=4)

_values = st.one_of(_float_arrays, _int_arrays, _bytes_lists)


@settings(max_examples=200, deadline=None)
@given(st.dictionaries(_names, _values, min_size=0, max_size=6))
def test_example_roundtrip(features):
  data = codec.encode_example(features)
  decoded = codec.decode_example(data)
  assert set(decoded) == set(features)
  for k, v in features.items():
    got = decoded[k]
    if isinstance(v, list):
      assert list(got) == v
    elif v.dtype == np.float32:
      np.testing.assert_array_equal(np.asarray(got, np.float32), v)
    else:
      np.testing.assert_array_equal(np.asarray(got, np.int64), v)


@settings(max_examples=100, deadline=None)
@given(
    st.dictionaries(_names, _values, min_size=0, max_size=3),
    st.dictionaries(
        _names,
        st.lists(_values, min_size=0, max_size=3), min_size=0, max_size=3))
def test_sequence_example_roundtrip(context, sequences):
  data = codec.encode_sequence_example(context, sequences)
  ctx, seqs = codec.decode_sequence_example(data)
  assert set(ctx) == set(context)
  assert set(seqs) == set(sequences)
  for k, steps in sequences.items():
    assert len(seqs[k]) == len(steps)


@settings(max_examples=60, deadline=None)
@given(st.lists(st.binary(min_size=0, max_size=300), min_size=0,
                max_size=10))
def test_tfrecord_roundtrip_property(tmp_path_factory, payloads):
  from tensor2robot_amd.data import tfrecord
  import os
  d = tmp_path_factory.mktemp("tfr")
  path = os.path.join(str(d), "r.tfrecord")
  with tfrecord.TFRecordWriter(path) as w:
    for p in payloads:
      w.write(p)
  got = list(tfrecord.read_records(path, verify_crc=True))
  assert got == payloads


def test_tfrecord_rejects_corrupt_crc(tmp_path):
  from tensor2robot_amd.data import tfrecord
  import os
  path = os.path.join(str(tmp_path), "r.tfrecord")
  with tfrecord.TFRecordWriter(path) as w:
    w.write(b"hello world")
  raw = bytearray(open(path, "rb").read())
  raw[-2] ^= 0xFF            # flip a bit inside the data CRC
  open(path, "wb").write(bytes(raw))
  import pytest as _pytest
  with _pytest.raises(IOError):
    list(tfrecord.read_records(path, verify_crc=True))


# ---- spec-structure flatten/pack property ----

_path_keys = st.lists(
    st.text(alphabet="abcdefgh", min_size=1, max_size=4),
    min_size=1, max_size=3).map(lambda parts: "/".join(parts))


@settings(max_examples=100, deadline=None)
@given(st.dictionaries(_path_keys, st.integers(0, 100),
                       min_size=1, max_size=8))
def test_spec_struct_flatten_pack_roundtrip(flat):
  """flatten(pack(flat)) == flat for any prefix-free path set
  (reference tensorspec_utils flatten/pack contract)."""
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  import torch
  # Drop keys where one path is a strict prefix of another (invalid
  # hierarchies: 'a' cannot be both a leaf and a subtree).
  keys = sorted(flat)
  pruned = {}
  for k in keys:
    if any(k != other and k.startswith(other + "/") for other in keys):
      continue
    pruned[k] = flat[k]
  keys = sorted(pruned)
  pruned = {k: v for k, v in pruned.items()
            if not any(k != o and o.startswith(k + "/") for o in keys)}
  if not pruned:
    return
  s = tsu.TensorSpecStruct()
  for k, v in pruned.items():
    s[k] = torch.tensor([v])
  flat_again = tsu.flatten_spec_structure(s)
  assert set(flat_again.keys()) == set(pruned.keys())
  for k, v in pruned.items():
    assert int(flat_again[k][0]) == v
