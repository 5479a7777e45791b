"""Space-to-depth stem equivalence + MIOpen DB utility (CPU)."""

import os

import torch
import torch.nn.functional as F

from tensor2robot_amd.ops import conv as conv_mod
from tensor2robot_amd.utils import miopen_db


def test_space_to_depth_conv_equivalence_cpu():
  """conv(x, 6x6/2 pad 2) == conv(s2d(x), s2d(w), 3x3/1 pad 1) exactly
  (fp32): the transformation behind the MFMA stem path."""
  torch.manual_seed(0)
  x = torch.randn(2, 3, 20, 24)
  w = torch.randn(8, 3, 6, 6)
  ref = F.conv2d(x, w, stride=2, padding=2)
  x2 = conv_mod._space_to_depth_nhwc(
      x.contiguous(memory_format=torch.channels_last))
  w2 = conv_mod._stem_weight_s2d(w.float())
  got = F.conv2d(x2, w2, stride=1, padding=1)
  torch.testing.assert_close(got, ref, atol=1e-4, rtol=1e-4)


def test_miopen_db_activation(tmp_path, monkeypatch):
  monkeypatch.delenv("MIOPEN_USER_DB_PATH", raising=False)
  # Packaged DB present -> copies to a writable dir and sets the env.
  assert os.path.isdir(miopen_db.DB_DIR) and os.listdir(miopen_db.DB_DIR)
  assert miopen_db.use_packaged_db()
  target = os.environ["MIOPEN_USER_DB_PATH"]
  assert os.path.isdir(target) and os.listdir(target)
  assert os.access(target, os.W_OK)
  # Caller-configured env wins (no-op).
  monkeypatch.setenv("MIOPEN_USER_DB_PATH", "/custom")
  assert miopen_db.use_packaged_db()
  assert os.environ["MIOPEN_USER_DB_PATH"] == "/custom"
