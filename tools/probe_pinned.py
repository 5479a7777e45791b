"""Probe pinned-memory write rates on the GPU box (staging design)."""
import concurrent.futures as cf
import sys
import time

import torch

x = torch.randint(0, 256, (32, 512, 640, 3), dtype=torch.uint8)
mb = x.numel() / 1e6
dev = torch.device("cuda:0")
torch.cuda.init()

pinned = torch.empty_like(x).pin_memory()
plain = torch.empty_like(x)


def bench(fn, n=10, warm=3):
  for _ in range(warm):
    fn()
  t0 = time.perf_counter()
  for _ in range(n):
    fn()
  return (time.perf_counter() - t0) / n


t = bench(lambda: plain.copy_(x))
print(f"plain copy_      {mb:6.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")
t = bench(lambda: pinned.copy_(x))
print(f"pinned copy_     {mb:6.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")

xf = x.reshape(-1)
pf = pinned.reshape(-1)
nthreads = 4
chunks = [(i * xf.numel() // nthreads, (i + 1) * xf.numel() // nthreads)
          for i in range(nthreads)]
pool = cf.ThreadPoolExecutor(nthreads)


def par_copy():
  futs = [pool.submit(lambda a, b: pf[a:b].copy_(xf[a:b]), a, b)
          for a, b in chunks]
  for f in futs:
    f.result()


t = bench(par_copy)
print(f"pinned 4-thread  {mb:6.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")

# H2D rates
t = bench(lambda: (x.to(dev), torch.cuda.synchronize()))
print(f"pageable H2D     {mb:6.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")
t = bench(lambda: (pinned.to(dev, non_blocking=True),
                   torch.cuda.synchronize()))
print(f"pinned H2D       {mb:6.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")
g = torch.empty(x.shape, dtype=x.dtype, device=dev)
t = bench(lambda: (g.copy_(pinned, non_blocking=True),
                   torch.cuda.synchronize()))
print(f"pinned H2D presized {mb:4.1f}MB {t*1000:7.3f} ms {mb/1e3/t:6.1f} GB/s")
