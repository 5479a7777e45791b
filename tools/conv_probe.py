"""Micro-benchmark: stem-conv variants to avoid MIOpen naive fallback.

The rocprof breakdown (profiles/) shows naive_conv_*_wrw consuming ~90% of
step time for the 6x6/2 C=3 stem.  Candidates:
  a) baseline: conv 6x6/2 pad2 C=3, channels_last
  b) same, NCHW contiguous
  c) channel-pad input 3->4, conv C=4
  d) space-to-depth(2): [N,12,236,236] then conv 3x3/1 pad1  (same math as
     6x6/2 on 2x2-blocked pixels; kernel 3x3 over 12ch = K 108)
  e) f32 conv for the stem only
"""

import time

import torch
import torch.nn.functional as F

torch.backends.cudnn.benchmark = True


def timeit(fn, iters=20, warmup=5):
  for _ in range(warmup):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters * 1000.0


def bench(name, x, w, stride, padding, dtype=torch.bfloat16, cl=True):
  x = x.to(dtype)
  w = w.to(dtype)
  if cl:
    x = x.contiguous(memory_format=torch.channels_last)
    w = w.contiguous(memory_format=torch.channels_last)
  else:
    x = x.contiguous()
    w = w.contiguous()
  x = x.requires_grad_(True)
  w = w.requires_grad_(True)

  def fwd_bwd():
    y = F.conv2d(x, w, stride=stride, padding=padding)
    y.sum().backward()
    x.grad = None
    w.grad = None

  ms = timeit(fwd_bwd)
  print(f"{name:40s} {ms:8.3f} ms/iter (fwd+bwd)")
  return ms


def main():
  N = 32
  dev = "cuda"
  g = torch.Generator(device="cpu").manual_seed(0)
  x3 = torch.randn(N, 3, 472, 472, generator=g).to(dev)
  w3 = torch.randn(64, 3, 6, 6, generator=g).to(dev)

  bench("a) 6x6/2 C=3 channels_last bf16", x3, w3, 2, 2)
  bench("b) 6x6/2 C=3 NCHW bf16", x3, w3, 2, 2, cl=False)

  x4 = torch.cat([x3, torch.zeros(N, 1, 472, 472, device=dev)], dim=1)
  w4 = torch.cat([w3, torch.zeros(64, 1, 6, 6, device=dev)], dim=1)
  bench("c) 6x6/2 C=4 channels_last bf16", x4, w4, 2, 2)

  # d) space-to-depth: x[N,3,472,472] -> [N,12,236,236]; w 6x6 -> 3x3x12.
  xs = x3.reshape(N, 3, 236, 2, 236, 2).permute(0, 1, 3, 5, 2, 4) \
      .reshape(N, 12, 236, 236)
  ws = w3.reshape(64, 3, 3, 2, 3, 2).permute(0, 1, 3, 5, 2, 4) \
      .reshape(64, 12, 3, 3)
  bench("d) s2d 3x3/1 C=12 channels_last bf16", xs, ws, 1, 1)

  bench("e) 6x6/2 C=3 channels_last f32", x3, w3, 2, 2,
        dtype=torch.float32)

  # Verify d) == a) numerically (f32).
  ya = F.conv2d(x3.float(), w3.float(), stride=2, padding=2)
  yd = F.conv2d(xs.float(), ws.float(), stride=1, padding=1)
  print("a vs d max diff:", (ya - yd).abs().max().item(),
        "shapes", ya.shape, yd.shape)

  # 5x5 SAME C=64 conv (block1 workhorse) for reference
  x64 = torch.randn(N, 64, 79, 79, generator=g).to(dev)
  w64 = torch.randn(64, 64, 5, 5, generator=g).to(dev)
  bench("f) 5x5/1 C=64 79x79 channels_last bf16", x64, w64, 1, 2)
  x64b = torch.randn(N, 64, 236, 236, generator=g).to(dev)
  bench("g) 5x5/1 C=64 236x236 channels_last bf16", x64b, w64, 1, 2)


if __name__ == "__main__":
  main()
