"""Probe rocBLAS/hipBLASLt on the GEMM-conv shapes (ResNet50 @ 472^2).

Hypothesis: the dw GEMMs (tiny output x huge contraction) run without
split-K and strand the chip; a chunked-bmm reduction fixes it.

  python tools/probe_gemm_shapes.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(fn, n=20, warm=5):
  for _ in range(warm):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(n):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / n


def tf(flops, t):
  return flops / t / 1e12


def main():
  dev = "cuda"
  shapes = [
      # (M, K, N) fwd conv GEMMs: col[M,K] @ w[K,N]
      (48 * 118 * 118, 576, 64),
      (48 * 59 * 59, 1152, 128),
      (48 * 30 * 30, 2304, 256),
      (48 * 15 * 15, 4608, 512),
  ]
  for m, k, n in shapes:
    a = torch.randn(m, k, device=dev).to(torch.bfloat16)
    b = torch.randn(k, n, device=dev).to(torch.bfloat16)
    t = bench(lambda: a @ b)
    fl = 2.0 * m * k * n
    # dw: a.t() @ dy  -> [K, N2] with contraction M
    dy = torch.randn(m, n, device=dev).to(torch.bfloat16)
    t_dw = bench(lambda: a.t() @ dy)
    # chunked-bmm dw: split M into 64 chunks, bmm + sum
    chunks = 64
    mc = m // chunks
    a3 = a[:chunks * mc].view(chunks, mc, k)
    d3 = dy[:chunks * mc].view(chunks, mc, n)
    t_dw_c = bench(
        lambda: torch.bmm(a3.transpose(1, 2), d3).sum(0))
    print(f"M={m:7d} K={k:5d} N={n:4d}: fwd {t*1e3:7.3f} ms "
          f"{tf(fl, t):6.0f} TF | dw {t_dw*1e3:7.3f} ms "
          f"{tf(fl, t_dw):6.0f} TF | dw-chunk {t_dw_c*1e3:7.3f} ms "
          f"{tf(fl, t_dw_c):6.0f} TF", flush=True)


if __name__ == "__main__":
  main()
