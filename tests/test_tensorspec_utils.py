"""Spec-system tests mirroring the reference's tensorspec_utils_test.py."""

import collections
import pickle

import numpy as np
import pytest
import torch

from tensor2robot_amd.specs import tensorspec_utils as tsu

T = tsu.ExtendedTensorSpec


def test_basic_spec_fields():
  s = T(shape=(224, 224, 3), dtype=torch.uint8, name="image",
        data_format="JPEG", dataset_key="d1")
  assert s.shape == (224, 224, 3)
  assert s.dtype == torch.uint8
  assert s.name == "image"
  assert s.data_format == "JPEG"
  assert s.dataset_key == "d1"
  assert not s.is_optional and not s.is_sequence


def test_dtype_coercion():
  assert T((3,), "float32").dtype == torch.float32
  assert T((3,), np.float32).dtype == torch.float32
  assert T((3,), torch.bfloat16).dtype == torch.bfloat16
  with pytest.raises(ValueError):
    T((3,), "notadtype")


def test_equality_shape_dtype_only():
  a = T((3,), torch.float32, name="a")
  b = T((3,), torch.float32, name="b", is_optional=True)
  c = T((4,), torch.float32)
  d = T((3,), torch.int32)
  assert a == b
  assert a != c and a != d


def test_from_spec_batch_size():
  s = T((4, 5), torch.float32, name="x")
  b = T.from_spec(s, batch_size=8)
  assert b.shape == (8, 4, 5)
  dyn = T.from_spec(s, batch_size=-1)
  assert dyn.shape == (None, 4, 5)
  unchanged = T.from_spec(s)
  assert unchanged.shape == (4, 5)
  assert unchanged.name == "x"


def test_from_tensor():
  t = torch.zeros(2, 3)
  s = T.from_tensor(t)
  assert s.shape == (2, 3) and s.dtype == torch.float32 and s.is_extracted
  s2 = T.from_tensor(np.zeros((5,), dtype=np.int64))
  assert s2.shape == (5,) and s2.dtype == torch.int64


def test_proto_dict_roundtrip():
  s = T((None, 3), torch.bfloat16, name="n", is_optional=True,
        is_sequence=True, data_format="PNG", dataset_key="k",
        varlen_default_value=None)
  d = s.to_proto_dict()
  s2 = T.from_proto_dict(d)
  assert s2.shape == (None, 3)
  assert s2.dtype == torch.bfloat16
  assert s2.name == "n" and s2.is_optional and s2.is_sequence
  assert s2.data_format == "PNG" and s2.dataset_key == "k"


def test_varlen_requires_shape():
  with pytest.raises(ValueError):
    T((), torch.float32, varlen_default_value=1.0)


# ---------------------------------------------------------------------------
# TensorSpecStruct
# ---------------------------------------------------------------------------


def _sample_struct():
  s = tsu.TensorSpecStruct()
  s["train/images"] = T((32, 32, 3), torch.float32, name="images")
  s["train/actions"] = T((8,), torch.float32, name="actions")
  s["eval/images"] = T((32, 32, 3), torch.float32, name="images")
  return s


def test_struct_flat_and_hierarchical():
  s = _sample_struct()
  assert list(s.keys()) == ["train/images", "train/actions", "eval/images"]
  sub = s.train
  assert isinstance(sub, tsu.TensorSpecStruct)
  assert list(sub.keys()) == ["images", "actions"]
  assert sub["images"] is s["train/images"]
  assert sub.images is s["train/images"]


def test_struct_views_are_live():
  s = _sample_struct()
  sub = s.train
  new = T((1,), torch.int32)
  sub["extra"] = new
  assert s["train/extra"] is new
  del s["train/extra"]
  assert "extra" not in sub


def test_struct_getitem_path_and_view():
  s = _sample_struct()
  assert s["train"]["images"] is s["train/images"]
  with pytest.raises(KeyError):
    _ = s["nope"]


def test_struct_set_nested_dict():
  s = tsu.TensorSpecStruct()
  s["cond"] = {"features": {"x": T((3,), torch.float32)}}
  assert list(s.keys()) == ["cond/features/x"]
  assert s.cond.features.x.shape == (3,)


def test_struct_delete_subtree():
  s = _sample_struct()
  del s["train"]
  assert list(s.keys()) == ["eval/images"]


def test_struct_invalid_keys():
  s = tsu.TensorSpecStruct()
  with pytest.raises(ValueError):
    s[""] = T((1,), torch.float32)
  with pytest.raises(ValueError):
    s["a//b"] = T((1,), torch.float32)


def test_struct_pickle():
  s = _sample_struct()
  s2 = pickle.loads(pickle.dumps(s))
  assert list(s2.keys()) == list(s.keys())
  assert s2["train/images"].shape == (32, 32, 3)


def test_struct_attribute_error():
  s = _sample_struct()
  with pytest.raises(AttributeError):
    _ = s.missing


# ---------------------------------------------------------------------------
# flatten / pack / validate
# ---------------------------------------------------------------------------


def test_flatten_nested_and_namedtuple():
  Pair = collections.namedtuple("Pair", ["train", "val"])
  spec = Pair(train={"x": T((2,), torch.float32)},
              val={"x": T((2,), torch.float32)})
  flat = tsu.flatten_spec_structure(spec)
  assert list(flat.keys()) == ["train/x", "val/x"]


def test_flatten_drops_none():
  flat = tsu.flatten_spec_structure(
      {"a": T((1,), torch.float32), "b": None})
  assert list(flat.keys()) == ["a"]


def test_pack_fills_and_raises():
  spec = tsu.TensorSpecStruct()
  spec["req"] = T((2,), torch.float32)
  spec["opt"] = T((2,), torch.float32, is_optional=True)
  packed = tsu.pack_flat_sequence_to_spec_structure(
      spec, {"req": np.zeros((2,), np.float32)})
  assert list(packed.keys()) == ["req"]
  with pytest.raises(ValueError):
    tsu.pack_flat_sequence_to_spec_structure(
        spec, {"opt": np.zeros((2,), np.float32)})


def test_validate_and_pack():
  spec = tsu.TensorSpecStruct()
  spec["x"] = T((None, 3), torch.float32)
  ok = {"x": torch.zeros(5, 3)}
  packed = tsu.validate_and_pack(spec, ok)
  assert packed["x"].shape == (5, 3)
  with pytest.raises(ValueError):
    tsu.validate_and_pack(spec, {"x": torch.zeros(5, 4)})
  with pytest.raises(ValueError):
    tsu.validate_and_pack(spec, {"x": torch.zeros(5, 3, dtype=torch.int32)})


def test_validate_ignore_batch():
  spec = {"x": T((7, 3), torch.float32)}
  tsu.assert_equal(spec, {"x": torch.zeros(9, 3)}, ignore_batch=True)
  with pytest.raises(ValueError):
    tsu.assert_equal(spec, {"x": torch.zeros(9, 3)}, ignore_batch=False)


def test_assert_valid_spec_structure_name_rule():
  good = {"a": T((2,), torch.float32, name="shared"),
          "b": T((2,), torch.float32, name="shared")}
  tsu.assert_valid_spec_structure(good)
  bad = {"a": T((2,), torch.float32, name="shared"),
         "b": T((3,), torch.float32, name="shared")}
  with pytest.raises(ValueError):
    tsu.assert_valid_spec_structure(bad)


def test_filter_by_dataset():
  spec = {"a": T((1,), torch.float32, dataset_key="d1"),
          "b": T((1,), torch.float32, dataset_key="d2"),
          "c": T((1,), torch.float32)}
  d1 = tsu.filter_spec_structure_by_dataset(spec, "d1")
  assert list(d1.keys()) == ["a"]
  default = tsu.filter_spec_structure_by_dataset(spec, "")
  assert list(default.keys()) == ["c"]


def test_copy_tensorspec_prefix_batch():
  spec = {"x": T((3,), torch.float32, name="x")}
  out = tsu.copy_tensorspec(spec, batch_size=4, prefix="cond")
  assert out["x"].shape == (4, 3)
  assert out["x"].name == "cond/x"


# ---------------------------------------------------------------------------
# factories
# ---------------------------------------------------------------------------


def test_make_random_numpy_shapes_dtypes():
  spec = {"img": T((4, 4, 3), torch.uint8),
          "vec": T((5,), torch.float32),
          "flag": T((), torch.bool)}
  out = tsu.make_random_numpy(spec, batch_size=2, seed=0)
  assert out["img"].shape == (2, 4, 4, 3) and out["img"].dtype == np.uint8
  assert out["vec"].shape == (2, 5) and out["vec"].dtype == np.float32
  assert out["flag"].shape == (2,)


def test_make_random_numpy_sequence():
  spec = {"seq": T((6,), torch.float32, is_sequence=True)}
  out = tsu.make_random_numpy(spec, batch_size=2, sequence_length=5)
  assert out["seq"].shape == (2, 5, 6)


def test_make_constant_numpy():
  spec = {"x": T((3,), torch.float32)}
  out = tsu.make_constant_numpy(spec, constant_value=2.5, batch_size=2)
  assert np.all(out["x"] == 2.5)


def test_make_random_tensors():
  spec = {"x": T((3,), torch.bfloat16)}
  out = tsu.make_random_tensors(spec, batch_size=2, seed=1)
  assert out["x"].dtype == torch.bfloat16 and out["x"].shape == (2, 3)


def test_pad_or_clip():
  spec = T((5, 2), torch.float32, varlen_default_value=3.0)
  short = torch.ones(2, 2)
  padded = tsu.pad_or_clip_tensor_to_spec_shape(short, spec)
  assert padded.shape == (5, 2)
  assert padded[2:].eq(3.0).all()
  long = torch.ones(9, 2)
  clipped = tsu.pad_or_clip_tensor_to_spec_shape(long, spec)
  assert clipped.shape == (5, 2)
  nppadded = tsu.pad_or_clip_tensor_to_spec_shape(
      np.ones((2, 2), np.float32), spec)
  assert nppadded.shape == (5, 2) and nppadded[3, 0] == 3.0


# ---------------------------------------------------------------------------
# dtype rewrites
# ---------------------------------------------------------------------------


def test_replace_dtype():
  spec = {"a": T((2,), torch.float32), "b": T((2,), torch.int32)}
  out = tsu.replace_dtype(spec, torch.float32, torch.bfloat16)
  assert out["a"].dtype == torch.bfloat16
  assert out["b"].dtype == torch.int32


def test_cast_f32_bf16_roundtrip():
  spec = {"a": T((2,), torch.bfloat16), "b": T((2,), torch.float32)}
  tensors = {"a": torch.zeros(2), "b": torch.zeros(2)}
  cast = tsu.cast_float32_to_bfloat16(tensors, spec)
  assert cast["a"].dtype == torch.bfloat16
  assert cast["b"].dtype == torch.float32
  back = tsu.cast_bfloat16_to_float32(cast)
  assert back["a"].dtype == torch.float32


# ---------------------------------------------------------------------------
# schema + assets
# ---------------------------------------------------------------------------


def test_tensorspec_to_feature_dict():
  spec = {"img": T((32, 32, 3), torch.uint8, name="image/encoded",
                   data_format="JPEG"),
          "seq": T((4,), torch.float32, name="s", is_sequence=True),
          "var": T((6, 2), torch.float32, name="v",
                   varlen_default_value=0.0)}
  schema, keys = tsu.tensorspec_to_feature_dict(spec)
  assert schema["image/encoded"].is_image
  assert schema["image/encoded"].kind == tsu.FeatureKind.FIXED_LEN
  assert schema["s"].kind == tsu.FeatureKind.FIXED_LEN_SEQUENCE
  assert schema["v"].kind == tsu.FeatureKind.VARLEN
  assert keys["image/encoded"] == ["img"]


def test_t2r_assets_roundtrip(tmp_path):
  feature_spec = tsu.TensorSpecStruct()
  feature_spec["state/img"] = T((8, 8, 3), torch.uint8, name="img",
                                data_format="JPEG")
  label_spec = tsu.TensorSpecStruct()
  label_spec["target"] = T((2,), torch.float32, name="target")
  assets = tsu.T2RAssets(feature_spec, label_spec, global_step=123)
  path = str(tmp_path / tsu.T2R_ASSETS_FILENAME)
  tsu.write_t2r_assets_to_file(assets, path)
  loaded = tsu.load_t2r_assets_from_file(path)
  assert loaded.global_step == 123
  assert loaded.feature_spec["state/img"].shape == (8, 8, 3)
  assert loaded.feature_spec["state/img"].data_format == "JPEG"
  assert loaded.label_spec["target"].dtype == torch.float32


def test_legacy_pkl_assets_roundtrip(tmp_path):
  import torch
  spec = tsu.TensorSpecStruct()
  spec["state/image"] = tsu.ExtendedTensorSpec((64, 64, 3), torch.uint8,
                                               name="img",
                                               data_format="jpeg")
  labels = tsu.TensorSpecStruct()
  labels["pose"] = tsu.ExtendedTensorSpec((2,), torch.float32, name="pose")
  pkl = str(tmp_path / "input_specifications.pkl")
  tsu.write_input_spec_to_pkl_file(pkl, spec, labels)
  f, l = tsu.load_input_spec_from_pkl_file(pkl)
  assert tuple(f["state/image"].shape) == (64, 64, 3)
  assert f["state/image"].data_format == "jpeg"
  assert l["pose"].dtype == torch.float32
  out = str(tmp_path / "t2r_assets.pbtxt")
  assets = tsu.convert_pkl_assets_to_proto_assets(pkl, out, global_step=7)
  assert assets.global_step == 7
  loaded = tsu.load_t2r_assets_from_file(out)
  assert "state/image" in loaded.feature_spec


def test_convert_pkl_assets_cli(tmp_path):
  import torch
  from tensor2robot_amd.bin import convert_pkl_assets
  spec = tsu.TensorSpecStruct()
  spec["obs"] = tsu.ExtendedTensorSpec((4,), torch.float32, name="obs")
  labels = tsu.TensorSpecStruct()
  labels["act"] = tsu.ExtendedTensorSpec((2,), torch.float32, name="act")
  assets_dir = tmp_path / "assets.extra"
  assets_dir.mkdir()
  tsu.write_input_spec_to_pkl_file(
      str(assets_dir / tsu.INPUT_SPEC_PKL_FILENAME), spec, labels)
  out = convert_pkl_assets.main(["--assets_filepath", str(assets_dir)])
  loaded = tsu.load_t2r_assets_from_file(out)
  assert "obs" in loaded.feature_spec and "act" in loaded.label_spec


def test_reference_named_api_parity(tmp_path):
  """Reference utils/tensorspec_utils.py entry-point names resolve and
  behave (convert/make_placeholders/from_tensors/lengths/feeds/IO)."""
  import torch
  spec = tsu.TensorSpecStruct()
  spec["obs/img"] = tsu.ExtendedTensorSpec((4, 4, 3), torch.uint8,
                                           name="img")
  spec["seq"] = tsu.ExtendedTensorSpec((2,), torch.float32, name="s",
                                       is_sequence=True)
  flat = tsu.convert_to_tensorspecstruct({"obs": {"img": spec["obs/img"]},
                                          "seq": spec["seq"]})
  assert "obs/img" in flat

  ph = tsu.make_placeholders(spec, batch_size=3, sequence_length=5)
  assert ph["obs/img"].shape == (3, 4, 4, 3)
  assert ph["seq"].shape == (3, 5, 2)
  assert tsu.make_placeholders(spec, batch_size=0)["obs/img"].shape == \
      (4, 4, 3)

  specs_back = tsu.tensorspec_from_tensors(ph)
  assert tuple(specs_back["obs/img"].shape) == (3, 4, 4, 3)

  with_len = tsu.add_sequence_length_specs(spec)
  assert "seq_length" in with_len
  assert with_len["seq_length"].dtype == torch.int64

  assert tsu.is_flat_spec_or_tensors_structure({"a": ph["seq"]})
  assert not tsu.is_flat_spec_or_tensors_structure({"a": {"b": 1}})

  np_in = tsu.make_random_numpy(spec, batch_size=3, sequence_length=5)
  fd = tsu.map_predict_fn_dict(spec, np_in)
  assert set(fd) == {"obs/img", "seq"}
  try:
    tsu.map_predict_fn_dict(spec, np_in, feed_dict=fd)
    assert False, "expected overwrite error"
  except ValueError:
    pass
  unsafe = tsu.map_feed_dict_unsafe(spec, np_in)
  assert set(unsafe) == {"obs/img", "seq"}

  # IO names.
  labels = tsu.TensorSpecStruct()
  labels["y"] = tsu.ExtendedTensorSpec((1,), torch.float32, name="y")
  pkl = str(tmp_path / "input_specifications.pkl")
  tsu.write_input_spec_to_file(spec, labels, pkl)
  f, l = tsu.load_input_spec_from_file(pkl)
  assert "obs/img" in f and "y" in l
  gs = str(tmp_path / "global_step.pkl")
  tsu.write_global_step_to_file(7, gs)
  assert tsu.load_global_step_from_file(gs) == 7
