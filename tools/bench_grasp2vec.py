"""Grasp2Vec training throughput on one MI355X (BASELINE config #4).

Full step through the Trainer fast path: three 472^2 image towers
(ResNet-50 spatial, shared scene tower over pre/post), NPairs loss,
backward + Adam, hipGraph-captured.

  python tools/bench_grasp2vec.py [--steps 30] [--warmup 15] [--batch-size 16]

DP=N (BASELINE config #4, one rank per MI355X over RCCL):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/bench_grasp2vec.py ...
"""

import argparse
import itertools
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tensor2robot_amd.models import optimizers
from tensor2robot_amd.research.grasp2vec import grasp2vec_model
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.train import train_eval


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--steps", type=int, default=30)
  p.add_argument("--warmup", type=int, default=15)
  p.add_argument("--batch-size", type=int, default=16)
  p.add_argument("--resnet-size", type=int, default=50)
  p.add_argument("--image-size", type=int, default=472)
  p.add_argument("--no-hipgraph", action="store_true")
  args = p.parse_args()
  import dist_bench
  distributed, rank, world_size, _dev = dist_bench.init()
  use_cuda = torch.cuda.is_available()

  sz = (args.image_size, args.image_size)
  model = grasp2vec_model.Grasp2VecModel(
      scene_size=sz, goal_size=sz, resnet_size=args.resnet_size,
      preprocessor_cls=None,
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-4),
      device_type="gpu" if use_cuda else "cpu",
      compute_dtype="bfloat16" if use_cuda else "float32")
  trainer = train_eval.Trainer(model, model_dir="",
                               use_hip_graph=use_cuda and
                               not args.no_hipgraph)
  device = trainer.device

  bs = args.batch_size
  g = torch.Generator().manual_seed(1234 + rank)
  pool = []
  for _ in range(2):
    f = tsu.TensorSpecStruct()
    for key in ("pregrasp_image", "postgrasp_image", "goal_image"):
      f[key] = torch.rand((bs,) + sz + (3,), generator=g).to(device)
    pool.append((f, tsu.TensorSpecStruct()))
  pool_iter = itertools.cycle(pool)

  def run_steps(n):
    trainer.train(lambda: pool_iter, trainer.global_step + n)

  run_steps(max(args.warmup, 20) if use_cuda else args.warmup)
  dist_bench.barrier_sync(distributed)
  if use_cuda:
    prev = None
    for _ in range(10):
      t0 = time.perf_counter()
      run_steps(5)
      dist_bench.barrier_sync(distributed)
      win = time.perf_counter() - t0
      stable = prev is not None and abs(win - prev) <= 0.05 * prev
      if distributed:
        import torch.distributed as dist
        flag = torch.tensor([1.0 if stable else 0.0], device=device)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        stable = float(flag.item()) >= 1.0
      if stable:
        break
      prev = win
  dist_bench.barrier_sync(distributed)
  t0 = time.perf_counter()
  run_steps(args.steps)
  dist_bench.barrier_sync(distributed)
  elapsed = dist_bench.max_over_ranks(time.perf_counter() - t0,
                                      distributed, device)
  if rank == 0:
    print(json.dumps({
        "metric": "images/sec (whole job, 3 towers) Grasp2Vec "
                  "ResNet%d train, %dx%d, bs=%d/GPU"
                  % (args.resnet_size, args.image_size,
                     args.image_size, bs),
        "value": round(bs * args.steps * world_size / elapsed, 2),
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "graphed": bool(trainer._fast_engine and
                        trainer._fast_engine.is_graphed),
        "dtype": "bf16" if use_cuda else "float32",
        "data": "synthetic", "n_gpus": world_size,
        "parallelism": "dp%d" % world_size,
    }))
  dist_bench.finalize(distributed)


if __name__ == "__main__":
  main()
