"""MFMA fragment-layout probe (guide rule: asymmetric operands)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
def test_mfma_32x32x16_bf16_layout():
  from tensor2robot_amd.ops import _t2r_hip
  torch.manual_seed(0)
  A = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
  B = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
  D = _t2r_hip.mfma_probe(A, B)
  ref = A.float() @ B.float()
  assert torch.allclose(D, ref, atol=2e-2, rtol=2e-2), \
      float((D - ref).abs().max())
  # Identity x asymmetric B: D == B rows exactly placed.
  I = torch.eye(32, 16, device="cuda").to(torch.bfloat16)
  B2 = torch.arange(16 * 32, device="cuda").reshape(16, 32).to(
      torch.bfloat16) / 100.0
  D2 = _t2r_hip.mfma_probe(I, B2)
  ref2 = I.float() @ B2.float()
  assert torch.allclose(D2, ref2, atol=2e-2, rtol=2e-2)
