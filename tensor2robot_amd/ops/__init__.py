"""Hand-written CDNA4 HIP kernel library (loud-fail loader).

On a GPU (ROCm) machine the extension MUST be present — ops raise rather
than silently falling back to eager torch, so a missing/broken build is
visible in benchmarks and GPU tests.  On CPU-only machines (CI) the torch
reference implementations run instead.
"""

from __future__ import annotations

import torch

import importlib

_t2r_hip = None
_load_error = None

try:
  _t2r_hip = importlib.import_module("tensor2robot_amd.ops._t2r_hip")
except ImportError as e:  # pragma: no cover
  _load_error = e


def hip_available() -> bool:
  return _t2r_hip is not None


def require_hip():
  """Returns the extension; raises loudly if we are on GPU without it."""
  if _t2r_hip is None:
    raise RuntimeError(
        "tensor2robot_amd HIP extension (_t2r_hip) is not built but a GPU "
        f"path was requested. Build with `python setup.py build_ext "
        f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). Import error: {_load_error}")
  return _t2r_hip


def hip_or_none():
  return _t2r_hip
