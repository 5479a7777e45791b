"""5x5 ring depth A/B (interleaved, same process/box).

  python tools/microbench_ring.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
  assert torch.cuda.is_available()
  from tensor2robot_amd.ops import _t2r_hip

  n, c, h, w, k, r, pad = 32, 64, 78, 78, 64, 5, 2
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  wt = torch.randn(k, c, r, r, device="cuda").to(torch.bfloat16)
  wpk = _t2r_hip.pack_conv_w(wt, False)

  # depth is latched per-process via a static; fork two child procs.
  import subprocess
  import json
  results = {}
  # in-process: run both depths by latching env BEFORE first dispatch is
  # not possible (static init), so compare via two alternating procs.
  for depth in ("6", "3", "w4", "3", "w4"):
    out = subprocess.run(
        [sys.executable, "-c", f"""
import os, sys, time
env_depth = "{depth}"\nif env_depth == "w4": os.environ["T2R_RING_WAVES"] = "4"\nelse: os.environ["T2R_RING_DEPTH"] = env_depth
sys.path.insert(0, {os.path.dirname(os.path.dirname(os.path.abspath(__file__)))!r})
import torch
from tensor2robot_amd.ops import _t2r_hip
n, c, h, w, k, r, pad = 32, 64, 78, 78, 64, 5, 2
x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16).contiguous(
    memory_format=torch.channels_last)
wt = torch.randn(k, c, r, r, device="cuda").to(torch.bfloat16)
wpk = _t2r_hip.pack_conv_w(wt, False)
y = _t2r_hip.conv_s1_nhwc(x, wpk, k, r, r, pad)
ref = torch.nn.functional.conv2d(x.float(), wt.float(), padding=pad)
err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
for _ in range(20):
  y = _t2r_hip.conv_s1_nhwc(x, wpk, k, r, r, pad)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(200):
  y = _t2r_hip.conv_s1_nhwc(x, wpk, k, r, r, pad)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / 200 * 1000
flops = 2.0 * n * h * w * r * r * c * k
print(f"{{ms:.4f}} {{flops / (ms / 1000) / 1e12:.1f}} relerr={{err:.2e}}")
"""], capture_output=True, text=True)
    line = out.stdout.strip().splitlines()[-1] if out.stdout.strip() else "fail"
    print(f"depth={depth}: {line}" + ("" if out.returncode == 0 else
                                      f"  [stderr: {out.stderr[-200:]}]"))


if __name__ == "__main__":
  main()
