"""BN grid-cap A/B on the big 78^2 layer (subprocess per cap)."""
import os, subprocess, sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CHILD = """
import os, sys, time
sys.path.insert(0, %r)
import torch
from tensor2robot_amd.ops import fused_bn
m = fused_bn.FusedBatchNormReLU(64).cuda().train()
x = torch.randn(32, 64, 78, 78, device="cuda").to(torch.bfloat16) \\
    .contiguous(memory_format=torch.channels_last).requires_grad_(True)
for _ in range(10):
  y = m(x); y.backward(y.detach())
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(100):
  y = m(x)
  y.backward(y.detach())
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / 100 * 1000
mb = 32*64*78*78*2 / 1e6
print(f"{ms:.4f} ms  eff_bytes={7*mb/ms:.0f} GB/s")
""" % (REPO,)


def main():
  for cap in ("1024", "2048", "4096", "1024", "2048", "4096"):
    env = dict(os.environ, T2R_BN_GRID_CAP=cap)
    out = subprocess.run([sys.executable, "-c", CHILD], env=env,
                         capture_output=True, text=True)
    line = out.stdout.strip().splitlines()[-1] if out.stdout.strip() else \
        "fail " + out.stderr[-150:]
    print(f"cap={cap}: {line}")


if __name__ == "__main__":
  main()
