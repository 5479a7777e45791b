"""GPU JPEG decode parity vs the CPU codec.

The HIP kernels (dequant + IDCT + upsample + color) must reproduce the
CPU baseline decode bit-near-exactly on images from our encoder and on
synthetic sampling variants.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


def _roundtrip_records(n, h, w, channels=3, quality=90, seed=0):
  from tensor2robot_amd.data import image_codec
  rng = np.random.RandomState(seed)
  records, refs = [], []
  for i in range(n):
    # Smooth-ish content (JPEG-friendly) + noise.
    yy, xx = np.mgrid[0:h, 0:w]
    base = (128 + 80 * np.sin(xx / (7.0 + i)) *
            np.cos(yy / (5.0 + i)))[..., None]
    img = np.clip(base + rng.randint(-30, 30, (h, w, channels)),
                  0, 255).astype(np.uint8)
    if channels == 1:
      img = img[:, :, 0]
    rec = image_codec.encode_jpeg(img, quality)
    records.append(rec)
    refs.append(image_codec.decode_jpeg(rec))
  return records, refs


@requires_gpu
@pytest.mark.parametrize("hw", [(64, 80), (100, 100), (57, 43)])
def test_gpu_jpeg_matches_cpu_codec(hw):
  from tensor2robot_amd.data import gpu_jpeg
  h, w = hw
  records, refs = _roundtrip_records(4, h, w)
  out = gpu_jpeg.decode_jpeg_batch(records, device="cuda")
  assert out.shape == (4, h, w, 3) and out.dtype == torch.uint8
  got = out.cpu().numpy()
  for i, ref in enumerate(refs):
    # IDCT rounding may differ by 1 LSB on a handful of pixels.
    diff = np.abs(got[i].astype(int) - ref.astype(int))
    assert diff.max() <= 1, (i, diff.max())
    assert (diff > 0).mean() < 0.02


@requires_gpu
def test_gpu_jpeg_grayscale():
  from tensor2robot_amd.data import gpu_jpeg
  records, refs = _roundtrip_records(3, 48, 56, channels=1)
  out = gpu_jpeg.decode_jpeg_batch(records, device="cuda")
  assert out.shape == (3, 48, 56, 1)
  got = out.cpu().numpy()[..., 0]
  for i, ref in enumerate(refs):
    diff = np.abs(got[i].astype(int) - ref.astype(int))
    assert diff.max() <= 1


@requires_gpu
def test_gpu_jpeg_mixed_quality_groups():
  """Different quant tables in one batch -> grouped dispatches."""
  from tensor2robot_amd.data import gpu_jpeg
  r1, refs1 = _roundtrip_records(2, 40, 40, quality=90, seed=1)
  r2, refs2 = _roundtrip_records(2, 40, 40, quality=60, seed=2)
  out = gpu_jpeg.decode_jpeg_batch(r1 + r2, device="cuda")
  got = out.cpu().numpy()
  for i, ref in enumerate(refs1 + refs2):
    diff = np.abs(got[i].astype(int) - ref.astype(int))
    assert diff.max() <= 1, (i, diff.max())


@requires_gpu
def test_parser_gpu_image_decode():
  """ExampleParser(image_decode_device='cuda') yields CUDA uint8
  batches decoded by the HIP path."""
  from tensor2robot_amd.data import example as example_mod
  from tensor2robot_amd.data import parser as parser_mod
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  records, refs = _roundtrip_records(3, 48, 64)
  spec = tsu.TensorSpecStruct()
  spec["img"] = tsu.ExtendedTensorSpec((48, 64, 3), torch.uint8,
                                       name="image", data_format="JPEG")
  recs = [example_mod.encode_example({"image": r}) for r in records]
  p = parser_mod.ExampleParser(spec, image_decode_device="cuda")
  out = p(recs)
  img = out["img"]
  assert img.is_cuda and img.dtype == torch.uint8
  got = img.cpu().numpy()
  for i, ref in enumerate(refs):
    assert np.abs(got[i].astype(int) - ref.astype(int)).max() <= 1
