"""BC-Z training throughput on one MI355X (BASELINE config #3 evidence).

Full step through the framework fast path (Trainer + FastStepEngine:
hipGraph-captured fwd+bwd+opt): on-GPU preprocess (crop 472^2 of
512x640 -> resize 100^2 -> distort) + FiLM-ResNet forward + component
losses + backward + Adam.

  python tools/bench_bcz.py [--steps 50] [--warmup 15] [--batch-size 32]

DP=N (BASELINE config #3, one rank per MI355X over RCCL):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/bench_bcz.py ...
"""

import argparse
import functools
import itertools
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tensor2robot_amd.models import optimizers
from tensor2robot_amd.research.bcz import model as bcz_model
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.train import train_eval
from tensor2robot_amd.utils import modes as run_modes


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--steps", type=int, default=50)
  p.add_argument("--warmup", type=int, default=15)
  p.add_argument("--batch-size", type=int, default=32)
  p.add_argument("--resnet-size", type=int, default=18)
  p.add_argument("--no-hipgraph", action="store_true")
  args = p.parse_args()
  import dist_bench
  distributed, rank, world_size, device0 = dist_bench.init()
  use_cuda = torch.cuda.is_available()

  model = bcz_model.BCZModel(
      image_size=(100, 100), input_size=(512, 640), num_waypoints=10,
      resnet_size=args.resnet_size,
      preprocessor_cls=functools.partial(
          bcz_model.BCZPreprocessor, image_size=(100, 100),
          input_size=(512, 640), crop_size=(472, 472), mock_subtask=True),
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
    features = tsu.TensorSpecStruct()
    features["image"] = torch.randint(0, 256, (bs, 512, 640, 3),
                                      dtype=torch.uint8,
                                      generator=g).to(device)
    features["subtask_id"] = torch.zeros(bs, 1, dtype=torch.int64,
                                         device=device)
    for name, size, _, _ in model._action_components:
      features["present/" + name] = torch.rand(bs, size,
                                               generator=g).to(device)
    labels = tsu.TensorSpecStruct()
    labels["future/xyz_residual"] = torch.randn(bs, 10, 3,
                                                generator=g).to(device)
    labels["future/quaternion"] = torch.randn(bs, 10, 4,
                                              generator=g).to(device)
    labels["future/target_close"] = (torch.rand(bs, 10, 1, generator=g)
                                     > 0.5).float().to(device)
    pool.append((features, labels))
  pool_iter = itertools.cycle(pool)
  preprocess_fn = functools.partial(model.preprocessor.preprocess,
                                    mode=run_modes.TRAIN)

  def run_steps(n):
    trainer.train(lambda: pool_iter, trainer.global_step + n,
                  preprocess_fn=preprocess_fn)

  run_steps(max(args.warmup, 20) if use_cuda else args.warmup)
  dist_bench.barrier_sync(distributed)
  if use_cuda:
    # settle probe (find/clock ramp); collective stop in DP mode
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
        "metric": "images/sec (whole job) BC-Z FiLM-ResNet%d train, "
                  "100x100 (512x640 raw), bs=%d/GPU"
                  % (args.resnet_size, bs),
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
