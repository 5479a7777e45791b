"""Fused GPU image preprocessing wrapper (crop + convert + distort).

Single-kernel-pair replacement for the torch op chain in
`preprocessors/distortion.py` when the raw uint8 batch is already on the
GPU (reference preprocessors/distortion.py:56-133 crop+distort and
image_transformations.py:176-265).  Distortion
parameter ranges match
`image_transformations.ApplyPhotometricImageDistortions` defaults
(brightness ±0.125, saturation [0.5,1.5], contrast [0.5,1.5]).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from tensor2robot_amd import ops as ops_mod
from tensor2robot_amd.preprocessors import image_transformations
from tensor2robot_amd.utils import modes as run_modes


def fused_preprocess_image(raw: torch.Tensor, mode: str,
                           target_shape: Tuple[int, int],
                           out_dtype: torch.dtype = torch.float32,
                           generator: Optional[torch.Generator] = None
                           ) -> torch.Tensor:
  """uint8 NHWC [N,H,W,3] on GPU -> cropped float [N,th,tw,3] in [0,1]."""
  ext = ops_mod.require_hip()
  n, h, w, c = raw.shape
  th, tw = target_shape
  if mode == run_modes.TRAIN:
    oy, ox = image_transformations.random_crop_offsets(
        (h, w), (th, tw), generator)
    delta_b = (torch.rand(n, device=raw.device) * 2.0 - 1.0) * 0.125
    f_sat = torch.rand(n, device=raw.device) + 0.5
    f_con = torch.rand(n, device=raw.device) + 0.5
    return ext.fused_preprocess(raw.contiguous(), oy, ox, th, tw,
                                delta_b, f_sat, f_con,
                                out_dtype == torch.bfloat16)
  oy, ox = (h - th) // 2, (w - tw) // 2
  return ext.fused_preprocess(raw.contiguous(), oy, ox, th, tw, None,
                              None, None, out_dtype == torch.bfloat16)
