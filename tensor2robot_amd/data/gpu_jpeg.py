"""Batched JPEG decode with the heavy stages on the GPU.

The MI355X-native replacement for the reference's in-pipeline
`tf.image.decode_image` (`utils/tfdata.py:426-484`, SURVEY §2.10 item
7).  The bit-serial Huffman scan runs on host threads (the C++ codec
releases the GIL, so a batch parallelizes across cores); dequant +
8x8 IDCT + chroma upsample + YCbCr->RGB run as two HIP kernels per
batch (ops/hip/jpeg_gpu.hip), writing the uint8 NHWC batch straight
into device memory where the fused preprocess kernel consumes it.

Images sharing geometry (the training case: fixed-shape datasets)
decode in ONE kernel pair per component; mixed geometries fall back to
per-image dispatches.
"""

from __future__ import annotations

import concurrent.futures
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from tensor2robot_amd import ops as ops_mod

_POOL: Optional[concurrent.futures.ThreadPoolExecutor] = None


def _pool() -> concurrent.futures.ThreadPoolExecutor:
  global _POOL
  if _POOL is None:
    import os
    _POOL = concurrent.futures.ThreadPoolExecutor(
        max_workers=min(16, os.cpu_count() or 4))
  return _POOL


def _huffman_batch(records: Sequence[bytes]) -> List[dict]:
  import os
  from tensor2robot_amd.data import image_codec
  native = image_codec.native_module()
  cores = os.cpu_count() or 4
  # Across-image threads come from the pool; when the batch is smaller
  # than the core count, spare cores decode restart-marker segments
  # WITHIN each image (no-op on streams without RST markers).
  per_image = max(1, cores // max(1, len(records)))
  if len(records) <= 1:
    return [native.decode_jpeg_coeffs(r, per_image) for r in records]
  return list(_pool().map(
      lambda r: native.decode_jpeg_coeffs(r, per_image), records))


def _geometry_key(ci: dict):
  return (ci["height"], ci["width"], ci["ncomp"],
          tuple((c["hs"], c["vs"]) for c in ci["comps"]),
          tuple(tuple(c["quant"].tolist()) for c in ci["comps"]))


def _decode_group(cis: List[dict], device) -> torch.Tensor:
  ext = ops_mod.require_hip()
  first = cis[0]
  h, w, ncomp = first["height"], first["width"], first["ncomp"]
  hmax, vmax = first["hmax"], first["vmax"]
  planes = []
  for c in range(ncomp):
    bh, bw, _ = first["comps"][c]["coeffs"].shape
    stacked = np.stack([ci["comps"][c]["coeffs"] for ci in cis])
    coeffs = torch.from_numpy(stacked).to(device, non_blocking=True)
    quant = torch.from_numpy(
        first["comps"][c]["quant"].astype(np.int32)).to(device)
    planes.append(ext.jpeg_idct(coeffs.reshape(-1, 64), quant, bh, bw))
  if ncomp == 1:
    return ext.jpeg_gray(planes[0], h, w)
  comps = first["comps"]
  return ext.jpeg_color(
      planes[0], planes[1], planes[2], h, w,
      comps[0]["hs"], comps[0]["vs"], comps[1]["hs"], comps[1]["vs"],
      hmax, vmax)


def decode_jpeg_batch(records: Sequence[bytes],
                      device="cuda") -> torch.Tensor:
  """[bytes] -> uint8 [N, H, W, C] on `device`.

  All records must share height/width (fixed-shape training data);
  mixed sampling/quant groups dispatch separately and re-interleave.
  """
  device = torch.device(device)
  cis = _huffman_batch(records)
  h, w = cis[0]["height"], cis[0]["width"]
  c_out = 3 if cis[0]["ncomp"] >= 3 else 1
  for ci in cis:
    if ci["height"] != h or ci["width"] != w:
      raise ValueError("decode_jpeg_batch: mixed image sizes")
  groups: Dict = {}
  for i, ci in enumerate(cis):
    groups.setdefault(_geometry_key(ci), []).append(i)
  if len(groups) == 1:
    return _decode_group(cis, device)
  out = torch.empty((len(cis), h, w, c_out), dtype=torch.uint8,
                    device=device)
  for idx_list in groups.values():
    dec = _decode_group([cis[i] for i in idx_list], device)
    out[torch.tensor(idx_list, device=device)] = dec
  return out
