"""In-pipeline JPEG re/compression of image tensors.

Reference `utils/tfdata.py:546-627` create_compress_fn /
create_decompress_fn: every spec with data_format == 'jpeg' is encoded
to (or decoded from) per-example JPEG bytes at quality 90 — the
reference's RAM-saving trick for replay buffers.  Uses the native
baseline JPEG codec (data/native/jpeg_codec.cpp).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.data import image_codec
from tensor2robot_amd.specs import tensorspec_utils as tsu


def _is_jpeg_spec(spec) -> bool:
  fmt = getattr(spec, "data_format", None)
  return fmt is not None and fmt.lower() == "jpeg"


def _to_uint8(tensor: torch.Tensor) -> np.ndarray:
  arr = tensor.detach().cpu()
  if arr.dtype.is_floating_point:
    arr = (arr.float().clamp(0.0, 1.0) * 255.0).round().to(torch.uint8)
  return arr.numpy()


@gin.configurable
def create_compress_fn(feature_spec, label_spec, quality: int = 90,
                       restart_rows: int = 1):
  """Returns compress_fn(features, labels) (reference :546-585).

  restart_rows > 0 writes an RSTn marker every that many MCU rows, so
  the decode-side Huffman scan parallelizes per segment
  (image_codec.encode_jpeg / gpu_jpeg) — free at encode time, ~4x
  faster entropy decode of big images on a 16-core host.
  """
  feature_spec = tsu.flatten_spec_structure(feature_spec)
  label_spec = tsu.flatten_spec_structure(label_spec) \
      if label_spec is not None else tsu.TensorSpecStruct()

  def compress_tensor(tensor: torch.Tensor) -> List[bytes]:
    batch = _to_uint8(tensor)
    interval = 0
    if restart_rows > 0 and len(batch):
      mcus_x = (batch[0].shape[1] + 7) // 8  # encoder is 4:4:4
      interval = mcus_x * restart_rows
    return [image_codec.encode_jpeg(img, quality=quality,
                                    restart_interval=interval)
            for img in batch]

  def compress_fn(features, labels=None):
    for key, spec in feature_spec.items():
      if _is_jpeg_spec(spec) and key in features:
        features[key] = compress_tensor(features[key])
    if labels is not None:
      for key, spec in label_spec.items():
        if _is_jpeg_spec(spec) and key in labels:
          labels[key] = compress_tensor(labels[key])
    return features, labels

  return compress_fn


@gin.configurable
def create_decompress_fn(feature_spec, label_spec):
  """Returns decompress_fn(features, labels) (reference :588-627)."""
  feature_spec = tsu.flatten_spec_structure(feature_spec)
  label_spec = tsu.flatten_spec_structure(label_spec) \
      if label_spec is not None else tsu.TensorSpecStruct()

  def decompress_tensor(data: List[bytes], spec) -> torch.Tensor:
    imgs = [image_codec.decode_jpeg(d) for d in data]
    batch = torch.from_numpy(np.stack(imgs))
    if spec.dtype.is_floating_point:
      batch = batch.to(spec.dtype) / 255.0
    shape = tuple(int(d) for d in spec.shape)
    return batch.reshape((len(data),) + shape)

  def decompress_fn(features, labels=None):
    for key, spec in feature_spec.items():
      if _is_jpeg_spec(spec) and key in features and \
          isinstance(features[key], (list, tuple)):
        features[key] = decompress_tensor(features[key], spec)
    if labels is not None:
      for key, spec in label_spec.items():
        if _is_jpeg_spec(spec) and key in labels and \
            isinstance(labels[key], (list, tuple)):
          labels[key] = decompress_tensor(labels[key], spec)
    return features, labels

  return decompress_fn
