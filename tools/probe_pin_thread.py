"""Is pinned memory allocated in a worker thread slower for H2D?"""
import sys, os, threading, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

torch.cuda.init()
dev = torch.device("cuda:0")
x = torch.randint(0, 256, (16, 512, 640, 3), dtype=torch.uint8)
mb = x.numel() / 1e6

def bench_h2d(buf, label, n=20):
  for _ in range(3):
    d = buf.to(dev, non_blocking=True)
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(n):
    d = buf.to(dev, non_blocking=True)
    torch.cuda.synchronize()
  t = (time.perf_counter() - t0) / n
  print(f"{label}: {t*1e3:7.3f} ms {mb/1e3/t:6.1f} GB/s pinned={buf.is_pinned()}")

main_pin = torch.empty_like(x).pin_memory(); main_pin.copy_(x)
bench_h2d(main_pin, "main-thread pinned ")

holder = {}
def worker():
  b = torch.empty_like(x).pin_memory()
  b.copy_(x)
  holder["buf"] = b
t = threading.Thread(target=worker); t.start(); t.join()
bench_h2d(holder["buf"], "worker-thread pinned")

bench_h2d(x, "pageable           ")

# H2D issued FROM the worker thread (ring producer pattern)
def worker2():
  bench_h2d(holder["buf"], "worker-issued H2D  ")
t2 = threading.Thread(target=worker2); t2.start(); t2.join()

# many small copies (the struct pattern): 9 tensors on one stream
small = [torch.rand(16, 8).pin_memory() for _ in range(9)]
for _ in range(3):
  for s in small: s.to(dev, non_blocking=True)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
  for s in small:
    s.to(dev, non_blocking=True)
  torch.cuda.synchronize()
print(f"9 small pinned copies: {(time.perf_counter()-t0)/20*1e3:.3f} ms/round")

# H2D concurrently with busy compute (the real pipeline condition).
a = torch.randn(8192, 8192, device=dev).to(torch.bfloat16)
b = torch.randn(8192, 8192, device=dev).to(torch.bfloat16)
comp = torch.cuda.Stream()
copy_s = torch.cuda.Stream()
dst = torch.empty(x.shape, dtype=x.dtype, device=dev)
torch.cuda.synchronize()
with torch.cuda.stream(comp):
  for _ in range(60):
    a @ b  # keep CUs + HBM busy ~hundreds of ms
t0 = time.perf_counter()
n = 20
with torch.cuda.stream(copy_s):
  for _ in range(n):
    dst.copy_(main_pin, non_blocking=True)
copy_s.synchronize()
t = (time.perf_counter() - t0) / n
print(f"H2D under GEMM load: {t*1e3:7.3f} ms {mb/1e3/t:6.1f} GB/s")
torch.cuda.synchronize()
