"""In-process A/B microbenchmarks (immune to box-to-box variance).

Times fused HIP ops against their torch counterparts, and the graphed
vs eager full train step, INTERLEAVED in one process so clocks/thermal
noise hits both sides equally.

Run on a GPU box:  python tools/microbench_ops.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, iters=50, warmup=10):
  for _ in range(warmup):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters * 1e3  # ms


def bench_maxpool():
  from tensor2robot_amd.ops import maxpool as fmp
  print("== maxpool fwd+bwd (ms), fused vs torch ==")
  for n, c, h, w, k in [(32, 64, 236, 236, 3), (32, 64, 79, 79, 3),
                        (32, 64, 27, 27, 2)]:
    x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    pool = fmp.FusedMaxPool2d(k, ceil_mode=True)
    dy = None

    def run_fused():
      xx = x.detach().requires_grad_(True)
      y = pool(xx)
      y.backward(torch.ones_like(y))

    def run_torch():
      xx = x.detach().requires_grad_(True)
      y = F.max_pool2d(xx, k, stride=k, ceil_mode=True)
      y.backward(torch.ones_like(y))

    # interleave
    t_f = t_t = 0.0
    for _ in range(3):
      t_f += timeit(run_fused, 20, 5)
      t_t += timeit(run_torch, 20, 5)
    print(f"  [{n},{c},{h},{w}] k={k}: fused {t_f/3:.3f}  "
          f"torch {t_t/3:.3f}  speedup {t_t/t_f:.2f}x")

    # fwd only
    t_ff = timeit(lambda: pool(x), 30, 5)
    t_tf = timeit(lambda: F.max_pool2d(x, k, stride=k, ceil_mode=True),
                  30, 5)
    print(f"      fwd only: fused {t_ff:.3f}  torch {t_tf:.3f}")


def bench_bn():
  from tensor2robot_amd.ops import fused_bn
  print("== fused BN+ReLU fwd+bwd (ms) vs torch BN+relu ==")
  for m, c in [(32 * 236 * 236, 64), (32 * 79 * 79, 64)]:
    x = torch.randn(m, c, device="cuda").to(torch.bfloat16)
    bn = fused_bn.FusedBatchNormReLU(c).cuda().train()
    tbn = torch.nn.BatchNorm1d(c, eps=1e-3, momentum=0.003).cuda() \
        .to(torch.float32).train()

    def run_fused():
      xx = x.detach().requires_grad_(True)
      y = bn(xx)
      y.backward(torch.ones_like(y))

    def run_torch():
      xx = x.detach().requires_grad_(True)
      y = torch.relu(tbn(xx.float()))
      y.backward(torch.ones_like(y))

    t_f = timeit(run_fused, 20, 5)
    t_t = timeit(run_torch, 20, 5)
    print(f"  [{m},{c}]: fused {t_f:.3f}  torch(f32) {t_t:.3f}")


def bench_full_step():
  """Eager vs hipGraph-captured QT-Opt train step, interleaved."""
  from tensor2robot_amd.models import optimizers as optimizers_mod
  from tensor2robot_amd.parallel import graph_step
  from tensor2robot_amd.research.qtopt import t2r_models

  device = torch.device("cuda:0")
  model = t2r_models.GraspingModel(device_type="gpu",
                                   compute_dtype="bfloat16",
                                   use_avg_model_params=True)
  model.to_device(device)
  network = model.network
  network.to(memory_format=torch.channels_last)
  optimizer = model.create_optimizer()
  ema = model.create_ema()
  bs = 32
  image = torch.randn(bs, 3, 472, 472, device=device).to(
      torch.bfloat16).contiguous(memory_format=torch.channels_last)
  action = torch.rand(bs, t2r_models.ACTION_DIM, device=device).to(
      torch.bfloat16)
  labels = (torch.rand(bs, device=device) > 0.5).float()
  autocast = torch.autocast("cuda", dtype=torch.bfloat16)

  def eager_step():
    optimizer.zero_grad(set_to_none=True)
    with autocast:
      logit = network(image, action)
    loss = F.binary_cross_entropy_with_logits(
        logit.float(), labels.reshape(logit.shape))
    loss.backward()
    optimizer.step(0)
    ema.update()
    return loss

  for _ in range(5):
    eager_step()
  torch.cuda.synchronize()
  graphed = graph_step.GraphedTrainStep(eager_step)

  t_e = t_g = 0.0
  for _ in range(3):
    t_e += timeit(eager_step, 30, 5)
    t_g += timeit(graphed.replay, 30, 5)
  print("== full train step (no preprocess), ms ==")
  print(f"  eager {t_e/3:.3f}  graphed {t_g/3:.3f}  "
        f"speedup {t_e/t_g:.2f}x")


def main():
  assert torch.cuda.is_available()
  torch.backends.cudnn.benchmark = True
  bench_maxpool()
  bench_bn()
  bench_full_step()


if __name__ == "__main__":
  main()
