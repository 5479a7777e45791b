"""A/B of T2R_RING_SETPRIO on the 5x5 ring conv (alternating procs).

s_setprio(1) around the global_load_lds issue phase is supposed to give
the memory-issuing wave priority over MFMA-bound waves (guide T4); the
flag is latched per-process, so each variant runs in a child process.

  python tools/microbench_setprio.py
"""

import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

_BODY = """
import os, sys, time
sys.path.insert(0, {root!r})
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
print(f"{{ms:.4f}} ms {{flops / (ms / 1000) / 1e12:.1f}} TF "
      f"relerr={{err:.2e}}")
"""


def main():
  root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
  for v in ("0", "1", "0", "1", "0", "1"):
    env = dict(os.environ)
    env["T2R_RING_SETPRIO"] = v
    out = subprocess.run([sys.executable, "-c", _BODY.format(root=root)],
                         capture_output=True, text=True, env=env)
    line = out.stdout.strip().splitlines()[-1] if out.stdout.strip() \
        else "fail"
    print(f"setprio={v}: {line}" +
          ("" if out.returncode == 0 else
           f"  [stderr: {out.stderr[-200:]}]"))


if __name__ == "__main__":
  main()
