"""Mode-dependent image preprocessing (reference preprocessors/distortion.py).

`preprocess_image` :56 — train: random crop + photometric distortion;
eval/predict: center crop.  `crop_image` :110 — the QT-Opt geometry
(512x640 -> 472x472 by default).  Sequence-aware via batch-dim folding.
"""

from __future__ import annotations

from typing import Optional

import torch

from tensor2robot_amd import gin
from tensor2robot_amd.preprocessors import image_transformations
from tensor2robot_amd.utils import modes as run_modes


@gin.configurable
def maybe_distort_image_batch(images: torch.Tensor, mode: str,
                              generator: Optional[torch.Generator] = None
                              ) -> torch.Tensor:
  """Photometric distortion in TRAIN mode only (reference :23-54)."""
  if mode == run_modes.TRAIN:
    (images,) = image_transformations.ApplyPhotometricImageDistortions(
        [images], random_brightness=True, random_saturation=True,
        random_hue=False, random_contrast=True, generator=generator)
  return images


@gin.configurable
def crop_image(img: torch.Tensor, mode: str,
               target_height: int = 472, target_width: int = 472,
               generator: Optional[torch.Generator] = None) -> torch.Tensor:
  """Random (train) or center (eval) crop; QT-Opt default 472x472 (:110)."""
  input_shape = (img.shape[1], img.shape[2])
  target_shape = (target_height, target_width)
  if mode == run_modes.TRAIN:
    (img,) = image_transformations.RandomCropImages(
        [img], input_shape, target_shape, generator=generator)
  else:
    (img,) = image_transformations.CenterCropImages(
        [img], input_shape, target_shape)
  return img


@gin.configurable
def preprocess_image(image: torch.Tensor, mode: str,
                     is_sequence: bool = False,
                     input_size=(512, 640), target_size=(472, 472),
                     crop_size=None,
                     generator: Optional[torch.Generator] = None
                     ) -> torch.Tensor:
  """uint8 NHWC -> float [0,1] -> crop -> distort(train) (reference :56-108).

  For sequences [N, T, H, W, C], time folds into batch for the transform.
  """
  leading = None
  if is_sequence or image.dim() == 5:
    leading = image.shape[:2]
    image = image.reshape(-1, *image.shape[2:])
  crop = crop_size or target_size
  needs_resize = tuple(crop) != tuple(target_size)
  if not needs_resize and image.is_cuda and image.dtype == torch.uint8 \
      and image.dim() == 4 and image.shape[-1] == 3:
    # Fused HIP path: crop + convert + photometric distortion in one
    # kernel pair (tensor2robot_amd/ops/hip/preprocess.hip).
    from tensor2robot_amd.ops import preprocess as fused
    image = fused.fused_preprocess_image(image, mode, tuple(crop),
                                         generator=generator)
    if leading is not None:
      image = image.reshape(*leading, *image.shape[1:])
    return image
  if image.dtype == torch.uint8:
    image = image.to(torch.float32) / 255.0
  image = crop_image(image, mode, crop[0], crop[1], generator=generator)
  if needs_resize:
    # Reference :96-97 resize_images (bilinear) after the crop.
    image = torch.nn.functional.interpolate(
        image.permute(0, 3, 1, 2), size=tuple(target_size),
        mode="bilinear", align_corners=False).permute(0, 2, 3, 1)
  image = maybe_distort_image_batch(image, mode, generator=generator)
  if leading is not None:
    image = image.reshape(*leading, *image.shape[1:])
  return image


def maybe_distort_and_flip_image_batch(images: torch.Tensor, mode: str):
  """TRAIN-only photometric distortion + random flips (reference
  distortion.py:39-53); 4D [B,H,W,C] or 5D [B,T,H,W,C]."""
  from tensor2robot_amd.utils import modes as run_modes
  if mode != run_modes.TRAIN:
    return images
  squeeze = False
  if images.dim() == 5:
    b, t = images.shape[:2]
    images = images.reshape(b * t, *images.shape[2:])
    squeeze = (b, t)
  images = image_transformations.ApplyPhotometricImageDistortions(
      [images])[0]
  images = image_transformations.ApplyRandomFlips([images])[0]
  if squeeze:
    b, t = squeeze
    images = images.reshape(b, t, *images.shape[1:])
  return images
