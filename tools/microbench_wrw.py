"""wrw v2 vs MIOpen interleaved microbench on the Grasping44 shapes.

  python tools/microbench_wrw.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tensor2robot_amd.utils import miopen_db

SHAPES = [
    # (n, c, h, w, k, r, pad)  -- the flagship wrw shapes
    (32, 64, 78, 78, 64, 5, 2),
    (32, 64, 26, 26, 64, 3, 1),
    (32, 64, 13, 13, 64, 3, 0),
]


def main():
  assert torch.cuda.is_available()
  miopen_db.use_packaged_db()
  torch.backends.cudnn.benchmark = True
  from tensor2robot_amd.ops import _t2r_hip

  for n, c, h, w, k, r, pad in SHAPES:
    x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = torch.randn(k, c, r, r, device="cuda").to(torch.bfloat16)
    oh, ow = h + 2 * pad - r + 1, w + 2 * pad - r + 1
    dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)

    def miopen():
      return torch.ops.aten.convolution_backward(
          dy, x, wt, None, (1, 1), (pad, pad), (1, 1), False, (0, 0), 1,
          (False, True, False))[1]

    def v2():
      return _t2r_hip.conv_s1_wrw2(x, dy, r, r, pad)

    def v3():
      return _t2r_hip.conv_s1_wrw3(x, dy, r, r, pad)

    def v4():
      return _t2r_hip.conv_s1_wrw4(x, dy, r, r, pad)

    for fn in (miopen, v2, v3, v4):      # warmup + find
      for _ in range(5):
        fn()
    torch.cuda.synchronize()
    times = {"miopen": 0.0, "v2": 0.0, "v3": 0.0, "v4": 0.0}
    iters = 50
    for _ in range(iters):           # interleaved
      for name, fn in (("miopen", miopen), ("v2", v2), ("v3", v3),
                       ("v4", v4)):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        times[name] += time.perf_counter() - t0
    flops = 2.0 * n * oh * ow * r * r * c * k
    for name in times:
      ms = times[name] / iters * 1000
      print(f"  {n}x{c}x{h}x{w} r={r}: {name:6s} {ms:7.3f} ms "
            f"{flops / (times[name] / iters) / 1e12:7.1f} TF")
    print(f"  ratio v2={times['miopen'] / times['v2']:.2f}x "
          f"v3={times['miopen'] / times['v3']:.2f}x "
          f"v4={times['miopen'] / times['v4']:.2f}x vs miopen")


if __name__ == "__main__":
  main()
