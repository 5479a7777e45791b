"""Bare BN fwd+bwd loop for rocprofv3 PMC attribution."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensor2robot_amd.ops import fused_bn

m = fused_bn.FusedBatchNormReLU(64).cuda().train()
x = torch.randn(32, 64, 78, 78, device="cuda").to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last).requires_grad_(True)
for _ in range(10):
  y = m(x); y.backward(y.detach())
torch.cuda.synchronize()
for _ in range(50):
  y = m(x); y.backward(y.detach())
torch.cuda.synchronize()
print("done")
