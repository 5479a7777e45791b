"""Grasp2Vec training throughput on one MI355X (BASELINE config #4).

Full step through the Trainer fast path: three 472^2 image towers
(ResNet-50 spatial, shared scene tower over pre/post), NPairs loss,
backward + Adam, hipGraph-captured.

  python tools/bench_grasp2vec.py [--steps 30] [--warmup 15] [--batch-size 16]
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
  assert torch.cuda.is_available()

  sz = (args.image_size, args.image_size)
  model = grasp2vec_model.Grasp2VecModel(
      scene_size=sz, goal_size=sz, resnet_size=args.resnet_size,
      preprocessor_cls=None,
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-4),
      device_type="gpu", compute_dtype="bfloat16")
  trainer = train_eval.Trainer(model, model_dir="",
                               use_hip_graph=not args.no_hipgraph)
  device = trainer.device

  bs = args.batch_size
  g = torch.Generator().manual_seed(0)
  pool = []
  for _ in range(2):
    f = tsu.TensorSpecStruct()
    for key in ("pregrasp_image", "postgrasp_image", "goal_image"):
      f[key] = torch.rand((bs,) + sz + (3,), generator=g).to(device)
    pool.append((f, tsu.TensorSpecStruct()))
  pool_iter = itertools.cycle(pool)

  def run_steps(n):
    trainer.train(lambda: pool_iter, trainer.global_step + n)

  run_steps(max(args.warmup, 20))
  torch.cuda.synchronize()
  prev = None
  for _ in range(10):
    t0 = time.perf_counter()
    run_steps(5)
    torch.cuda.synchronize()
    win = time.perf_counter() - t0
    if prev is not None and abs(win - prev) <= 0.05 * prev:
      break
    prev = win
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  run_steps(args.steps)
  torch.cuda.synchronize()
  elapsed = time.perf_counter() - t0
  print(json.dumps({
      "metric": "images/sec (3 towers) Grasp2Vec ResNet%d train, "
                "%dx%d, bs=%d" % (args.resnet_size, args.image_size,
                                  args.image_size, bs),
      "value": round(bs * args.steps / elapsed, 2),
      "ms_per_step": round(elapsed / args.steps * 1000, 3),
      "graphed": bool(trainer._fast_engine and
                      trainer._fast_engine.is_graphed),
      "dtype": "bf16", "data": "synthetic", "n_gpus": 1,
  }))


if __name__ == "__main__":
  main()
