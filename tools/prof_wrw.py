"""Bare wrw kernel loop for rocprofv3 attribution."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensor2robot_amd.ops import _t2r_hip

n, c, h, w, k, r, pad = 32, 64, 78, 78, 64, 5, 2
x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
oh = ow = 78
dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
for _ in range(5):
  _t2r_hip.conv_s1_wrw3(x, dy, r, r, pad)
  _t2r_hip.conv_s1_wrw2(x, dy, r, r, pad)
torch.cuda.synchronize()
for _ in range(50):
  _t2r_hip.conv_s1_wrw3(x, dy, r, r, pad)
torch.cuda.synchronize()
for _ in range(50):
  _t2r_hip.conv_s1_wrw2(x, dy, r, r, pad)
torch.cuda.synchronize()
print("done")
