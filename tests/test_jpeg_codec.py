"""Native baseline JPEG codec tests (cross-validated against PIL)."""

import io

import numpy as np
import pytest

from tensor2robot_amd.data import image_codec

PIL_Image = pytest.importorskip("PIL.Image")


def _gradient(h=96, w=128):
  y, x = np.mgrid[0:h, 0:w]
  return np.stack([(x * 255 / w), (y * 255 / h),
                   ((x + y) * 255 / (h + w))], axis=-1).astype(np.uint8)


def test_roundtrip_self():
  img = _gradient()
  data = image_codec.encode_jpeg(img, quality=95)
  dec = image_codec.decode_jpeg(data)
  assert dec.shape == img.shape
  assert np.abs(dec.astype(int) - img.astype(int)).mean() < 2.0


def test_pil_decodes_our_jpeg():
  img = _gradient()
  data = image_codec.encode_jpeg(img, quality=95)
  pil = np.asarray(PIL_Image.open(io.BytesIO(data)).convert("RGB"))
  assert pil.shape == img.shape
  assert np.abs(pil.astype(int) - img.astype(int)).mean() < 2.0


@pytest.mark.parametrize("subsampling", [0, 1, 2])
def test_we_decode_pil_jpeg(subsampling):
  """PIL emits 4:4:4 / 4:2:2 / 4:2:0 for subsampling 0/1/2."""
  img = _gradient(70, 90)  # odd-ish sizes exercise edge MCUs
  buf = io.BytesIO()
  PIL_Image.fromarray(img).save(buf, format="JPEG", quality=95,
                                subsampling=subsampling)
  dec = image_codec.decode_jpeg(buf.getvalue())
  pil_dec = np.asarray(PIL_Image.open(io.BytesIO(buf.getvalue()))
                       .convert("RGB"))
  assert dec.shape == img.shape
  # Our decode agrees with PIL's decode of the same file.
  assert np.abs(dec.astype(int) - pil_dec.astype(int)).mean() < 2.0


def test_we_decode_pil_grayscale_and_restart_markers():
  img = _gradient(64, 64)[:, :, 0]
  buf = io.BytesIO()
  PIL_Image.fromarray(img, mode="L").save(buf, format="JPEG",
                                          quality=90, restart_marker_rows=2)
  dec = image_codec.decode_jpeg(buf.getvalue())
  assert dec.shape == img.shape
  pil_dec = np.asarray(PIL_Image.open(io.BytesIO(buf.getvalue())))
  assert np.abs(dec.astype(int) - pil_dec.astype(int)).mean() < 1.5


def test_restart_interval_encode_roundtrip_and_pil():
  """Our encoder's RSTn streams decode identically and conform (PIL)."""
  img = _gradient(70, 90)  # edge MCUs + restart boundaries interact
  plain = image_codec.encode_jpeg(img, quality=95)
  rst = image_codec.encode_jpeg(img, quality=95, restart_interval=5)
  assert b"\xff\xdd" in rst and b"\xff\xdd" not in plain  # DRI present
  # Identical pixels either way (same quantization, only DC-pred resets).
  np.testing.assert_array_equal(image_codec.decode_jpeg(plain),
                                image_codec.decode_jpeg(rst))
  pil = np.asarray(PIL_Image.open(io.BytesIO(rst)).convert("RGB"))
  assert np.abs(pil.astype(int) - img.astype(int)).mean() < 2.0


@pytest.mark.parametrize("interval,threads", [(3, 4), (7, 16), (1, 2)])
def test_parallel_segment_coeff_decode_matches(interval, threads):
  """Segment-parallel Huffman == sequential scan, coefficient-exact."""
  native = image_codec.native_module()
  img = _gradient(56, 72)
  data = image_codec.encode_jpeg(img, quality=90,
                                 restart_interval=interval)
  seq = native.decode_jpeg_coeffs(data, 1)
  par = native.decode_jpeg_coeffs(data, threads)
  assert seq["height"] == par["height"] and seq["width"] == par["width"]
  for cs, cp in zip(seq["comps"], par["comps"]):
    np.testing.assert_array_equal(cs["coeffs"], cp["coeffs"])
  # num_threads on a marker-free stream is a no-op.
  plain = image_codec.encode_jpeg(img, quality=90)
  a = native.decode_jpeg_coeffs(plain, threads)
  b = native.decode_jpeg_coeffs(plain, 1)
  for cs, cp in zip(a["comps"], b["comps"]):
    np.testing.assert_array_equal(cs["coeffs"], cp["coeffs"])


def test_parallel_segment_decode_of_pil_restart_stream():
  """Foreign (PIL-encoded) restart streams hit the parallel path too."""
  native = image_codec.native_module()
  img = _gradient(64, 64)
  buf = io.BytesIO()
  PIL_Image.fromarray(img).save(buf, format="JPEG", quality=92,
                                subsampling=0, restart_marker_rows=1)
  seq = native.decode_jpeg_coeffs(buf.getvalue(), 1)
  par = native.decode_jpeg_coeffs(buf.getvalue(), 8)
  for cs, cp in zip(seq["comps"], par["comps"]):
    np.testing.assert_array_equal(cs["coeffs"], cp["coeffs"])


def test_decode_image_sniffs_jpeg_and_png():
  img = _gradient(32, 32)
  jp = image_codec.encode_jpeg(img, 90)
  assert image_codec.decode_image(jp).shape == img.shape
  png = image_codec.encode_png(img)
  np.testing.assert_array_equal(image_codec.decode_image(png), img)


def test_compress_decompress_fns():
  import torch
  from tensor2robot_amd.data import compression
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  spec = tsu.TensorSpecStruct()
  spec["image"] = tsu.ExtendedTensorSpec((32, 32, 3), torch.float32,
                                         name="img", data_format="jpeg")
  spec["pose"] = tsu.ExtendedTensorSpec((2,), torch.float32, name="pose")
  features = tsu.TensorSpecStruct()
  img = torch.from_numpy(_gradient(32, 32)).float() / 255.0
  features["image"] = img.unsqueeze(0).repeat(3, 1, 1, 1)
  features["pose"] = torch.zeros(3, 2)
  compress = compression.create_compress_fn(spec, None)
  f, _ = compress(features)
  assert isinstance(f["image"], list) and len(f["image"]) == 3
  assert isinstance(f["pose"], torch.Tensor)  # untouched
  decompress = compression.create_decompress_fn(spec, None)
  f2, _ = decompress(f)
  assert f2["image"].shape == (3, 32, 32, 3)
  err = (f2["image"] - features["image"]).abs().mean()
  assert float(err) < 0.02, float(err)
