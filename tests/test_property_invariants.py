"""Property-based invariants (hypothesis): wire codec, TFRecord
framing, spec-structure flatten/pack, crop/distortion bounds.

These sweep dtypes, shapes, extreme values and unicode names far past
the hand-picked cases in the per-module suites.
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


# ---- image transformation invariants ----

@settings(max_examples=50, deadline=None)
@given(st.integers(4, 16), st.integers(4, 16), st.integers(1, 3),
       st.integers(0, 2**31 - 1))
def test_photometric_distortions_stay_in_range(h, w, n, seed):
  """Reference image_transformations.py:176-265: distorted images are
  clipped to [0, 1] for any input and random draw."""
  import torch
  from tensor2robot_amd.preprocessors import image_transformations
  torch.manual_seed(seed)
  imgs = [torch.rand(2, h, w, 3) * 1.5 for _ in range(n)]  # some > 1
  out = image_transformations.ApplyPhotometricImageDistortions(
      [i.clamp(0, 1) for i in imgs], random_brightness=True,
      random_saturation=True, random_contrast=True,
      random_noise_levels=0.1, random_noise_apply_probability=1.0)
  for o in out:
    assert float(o.min()) >= 0.0 and float(o.max()) <= 1.0
    assert o.shape == (2, h, w, 3)


@settings(max_examples=50, deadline=None)
@given(st.integers(6, 20), st.integers(6, 20), st.integers(1, 6),
       st.integers(1, 6), st.integers(0, 2**31 - 1))
def test_random_crop_shares_offset_and_bounds(h, w, dh, dw, seed):
  """Reference :25-60: ONE random offset shared across the image list,
  output exactly the target shape, contents a true sub-window."""
  import torch
  from tensor2robot_amd.preprocessors import image_transformations
  ch, cw = max(1, h - dh), max(1, w - dw)
  torch.manual_seed(seed)
  base = torch.arange(h * w, dtype=torch.float32).reshape(1, h, w, 1)
  base = base.expand(2, h, w, 3).contiguous()
  a, b = image_transformations.RandomCropImages(
      [base, base.clone()], (h, w), (ch, cw))
  assert a.shape == (2, ch, cw, 3)
  torch.testing.assert_close(a, b)   # same offset for the whole list
  # The crop is a contiguous sub-window: top-left value determines all.
  tl = a[0, 0, 0, 0]
  row0 = int(tl) // w
  col0 = int(tl) % w
  assert 0 <= row0 <= h - ch and 0 <= col0 <= w - cw
  torch.testing.assert_close(
      a[0, :, :, 0], base[0, row0:row0 + ch, col0:col0 + cw, 0])
