"""CEM serving latency on one MI355X (reference README:53-58 contract).

Measures end-to-end SelectAction latency of the QT-Opt CEM policy: per
action, cem_iterations x one megabatch-tiled critic forward (the image
tower runs ONCE per iteration; the embedding is tiled across the action
samples, reference networks.py:515-521).  The reference's design target
is 1-10 Hz serving on a robot workstation; this reports the MI355X
number.

  python tools/bench_cem.py [--samples 64] [--iters 3] [--actions 50]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from tensor2robot_amd.utils import miopen_db

from tensor2robot_amd.predictors import checkpoint_predictor
from tensor2robot_amd.research.qtopt import t2r_models


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--samples", type=int, default=64)
  p.add_argument("--iters", type=int, default=3)
  p.add_argument("--actions", type=int, default=50)
  p.add_argument("--warmup", type=int, default=10)
  args = p.parse_args()
  assert torch.cuda.is_available()
  miopen_db.use_packaged_db()
  torch.backends.cudnn.benchmark = True

  model = t2r_models.GraspingModel(
      device_type="gpu", compute_dtype="bfloat16",
      action_batch_size=args.samples)
  predictor = checkpoint_predictor.CheckpointPredictor(
      t2r_model=model, device="cuda:0")
  predictor.init_randomly()
  policy = t2r_models.GraspingCEMPolicy(
      predictor=predictor, cem_samples=args.samples,
      cem_iterations=args.iters, seed=0)

  rng = np.random.RandomState(0)
  state = rng.randint(0, 256, (t2r_models.RAW_HEIGHT,
                               t2r_models.RAW_WIDTH, 3)).astype(np.uint8)

  for _ in range(args.warmup):
    policy.SelectAction(state)
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(args.actions):
    action = policy.SelectAction(state)
  torch.cuda.synchronize()
  elapsed = time.perf_counter() - t0
  ms = elapsed / args.actions * 1000
  print(json.dumps({
      "metric": "CEM SelectAction latency, QT-Opt critic, "
                f"{args.samples} samples x {args.iters} iters",
      "ms_per_action": round(ms, 3),
      "actions_per_sec": round(1000.0 / ms, 2),
      "action_dim": int(np.asarray(action).size),
      "dtype": "bf16", "n_gpus": 1,
  }))


if __name__ == "__main__":
  main()
