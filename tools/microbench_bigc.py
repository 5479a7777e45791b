"""A/B: C-chunked MFMA 3x3 (conv_s1_big.hip) vs GEMM-conv path.

ResNet-family stride-1 3x3 shapes at Grasp2Vec/BC-Z sizes; forward and
forward+backward timed separately.

  python tools/microbench_bigc.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch


SHAPES = [
    # n, c, h, w, k   (pad 1)
    (32, 128, 59, 59, 128),
    (32, 256, 30, 30, 256),
    (32, 512, 15, 15, 512),
    (32, 64, 118, 118, 64),     # layer1 (base conv_s1 territory)
    (32, 128, 25, 25, 128),     # BC-Z @100^2
]


def time_fn(fn, iters=100):
  for _ in range(15):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters


def main():
  assert torch.cuda.is_available()
  from tensor2robot_amd.ops import conv as conv_mod
  from tensor2robot_amd.ops import gemm_conv
  torch.manual_seed(0)
  for n, c, h, w, k in SHAPES:
    x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device="cuda") * 0.05).to(
        torch.bfloat16)
    flops_fwd = 2.0 * n * h * w * 9 * c * k  # pad-1 SAME
    bigc_ok = c % 32 == 0 and k % 64 == 0 and c >= 96

    # ---- forward only ----
    res = {}
    if bigc_ok:
      res["bigc"] = time_fn(
          lambda: conv_mod._BigCConvFunction.apply(x, wt, 1))
    res["gemm"] = time_fn(
        lambda: gemm_conv.gemm_conv2d(x, wt, (1, 1), (1, 1)))
    line = f"{n}x{c}x{h}x{w} k={k} fwd:"
    for name, dt in res.items():
      line += f"  {name} {dt*1e3:7.3f} ms {flops_fwd/dt/1e12:6.1f} TF"
    print(line, flush=True)

    # ---- fwd+bwd (dx + dw) ----
    def fb(fn):
      xg = x.detach().requires_grad_(True)
      wg = wt.detach().requires_grad_(True)
      y = fn(xg, wg)
      y.backward(torch.ones_like(y))
    res = {}
    if bigc_ok and c % 64 == 0:
      res["bigc"] = time_fn(
          lambda: fb(lambda a, b:
                     conv_mod._BigCConvFunction.apply(a, b, 1)), 50)
    res["gemm"] = time_fn(
        lambda: fb(lambda a, b:
                   gemm_conv.gemm_conv2d(a, b, (1, 1), (1, 1))), 50)
    line = f"{n}x{c}x{h}x{w} k={k} f+b:"
    for name, dt in res.items():
      line += f"  {name} {dt*1e3:7.3f} ms {3*flops_fwd/dt/1e12:6.1f} TF"
    print(line, flush=True)


if __name__ == "__main__":
  main()
