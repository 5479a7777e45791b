"""Streaming-rate probe: bn_apply vs plain copies across sizes.

Finds the achievable stream rate for the BN access pattern (16 B/lane
uint4, read-modify-write) at L3-resident and HBM-spilling sizes, vs
torch's contiguous copy as the device ceiling.

  python tools/probe_stream.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tensor2robot_amd.ops import _t2r_hip  # noqa: F401  (built .so)
from tensor2robot_amd import ops as ops_mod


def bench(fn, n=30, warm=5):
  for _ in range(warm):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(n):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / n


def main():
  ext = ops_mod.require_hip()
  C = 64
  for m_rows in (194688, 778752, 3115008, 12460032):
    # 194688 = the 32x78x78 flagship layer (24.9 MB); x4 steps up to
    # 1.6 GB (far past L3).
    mb = m_rows * C * 2 / 1e6
    x = torch.randn(m_rows, C, device="cuda").to(torch.bfloat16)
    scale = torch.ones(C, device="cuda")
    shift = torch.zeros(C, device="cuda")
    y = ext.bn_inference_apply(x, scale, shift, True)

    t = bench(lambda: ext.bn_inference_apply(x, scale, shift, True))
    rate_apply = 2 * mb / 1e3 / t  # read + write
    # ceiling reference: contiguous bf16 copy (read + write)
    dst = torch.empty_like(x)
    t2 = bench(lambda: dst.copy_(x))
    rate_copy = 2 * mb / 1e3 / t2
    # reduction-only reference (bn_stats-like read stream): torch sum
    t3 = bench(lambda: x.float().sum())  # not pure; use bn stats below
    print(f"M={m_rows:9d} ({mb:7.1f} MB): apply {t*1e3:7.3f} ms "
          f"{rate_apply:5.2f} TB/s | copy {t2*1e3:7.3f} ms "
          f"{rate_copy:5.2f} TB/s", flush=True)


if __name__ == "__main__":
  main()
