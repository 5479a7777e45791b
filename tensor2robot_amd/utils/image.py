"""Image <-> serialized-string helpers for replay writers.

Reference `utils/image.py`: jpeg_string :24, numpy_to_image_string :44.
Backed by the native codecs (data/image_codec) instead of PIL.
"""

from __future__ import annotations

import numpy as np

from tensor2robot_amd.data import image_codec


def jpeg_string(image: np.ndarray, jpeg_quality: int = 90) -> bytes:
  """uint8 HxWx3 (or HxW) array -> serialized JPEG (reference :24)."""
  return image_codec.encode_jpeg(np.asarray(image, np.uint8),
                                 quality=jpeg_quality)


def numpy_to_image_string(image_array: np.ndarray,
                          image_format: str = "jpeg",
                          data_type=np.uint8) -> bytes:
  """Array -> serialized image of the given format (reference :44)."""
  arr = np.asarray(image_array).astype(data_type)
  return image_codec.encode_image(arr, image_format)
