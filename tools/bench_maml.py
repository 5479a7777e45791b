"""Loop-vs-vmap A/B for the MAML inner loop (BASELINE config #5 shape).

Times the full outer step (MAMLModel.model_fn forward incl. per-task
second-order adaptation, outer backward, Adam) for the Python per-task
loop vs the torch.func.vmap task-parallel path
(`maml_inner_loop.inner_loop_vmapped`, reference use_parallel_for).

The workload mirrors Watch-Try-Learn statespace trial training: an MLP
policy adapted per task from condition episodes (reference
`research/vrgripper/configs/run_train_wtl_statespace_trial.gin`).

  python tools/bench_maml.py --tasks 16 --samples 40 --steps 30

DP=N (BASELINE config #5 shape, one rank per MI355X over RCCL;
gradients all-reduced after the outer backward):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/bench_maml.py ...
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
from torch import nn

from tensor2robot_amd.meta_learning import maml_model
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes


class _StatePolicy(abstract_model.AbstractT2RModel):
  """WTL-statespace-like MLP policy: state -> action (mse)."""

  def __init__(self, state_dim=32, action_dim=7, hidden=256, **kwargs):
    kwargs.setdefault("compute_dtype", "float32")
    super().__init__(**kwargs)
    self._state_dim = state_dim
    self._action_dim = action_dim
    self._hidden = hidden

  def get_feature_specification(self, mode):
    s = tsu.TensorSpecStruct()
    s["state"] = tsu.ExtendedTensorSpec((self._state_dim,),
                                        torch.float32, name="state")
    return s

  def get_label_specification(self, mode):
    s = tsu.TensorSpecStruct()
    s["action"] = tsu.ExtendedTensorSpec((self._action_dim,),
                                         torch.float32, name="action")
    return s

  def create_network(self):
    h = self._hidden
    return nn.Sequential(
        nn.Linear(self._state_dim, h), nn.ReLU(),
        nn.Linear(h, h), nn.ReLU(),
        nn.Linear(h, self._action_dim))

  def inference_network_fn(self, features, labels, mode, params=None):
    return {"inference_output": self.network(features["state"])}

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    return torch.nn.functional.mse_loss(
        inference_outputs["inference_output"], labels["action"])


class _BenchMAML(maml_model.MAMLModel):

  def _select_inference_output(self, predictions):
    predictions["condition_output"] = predictions[
        "full_condition_output/inference_output"]
    predictions["inference_output"] = predictions[
        "full_inference_output/inference_output"]
    return predictions


def _meta_batch(tasks, samples, state_dim, action_dim, device, seed=0):
  g = torch.Generator().manual_seed(seed)
  f = tsu.TensorSpecStruct()
  f["condition/features/state"] = torch.randn(
      tasks, samples, state_dim, generator=g).to(device)
  f["condition/labels/action"] = torch.randn(
      tasks, samples, action_dim, generator=g).to(device)
  f["inference/features/state"] = torch.randn(
      tasks, samples, state_dim, generator=g).to(device)
  l = tsu.TensorSpecStruct()
  l["action"] = torch.randn(tasks, samples, action_dim,
                            generator=g).to(device)
  return f, l


def bench(parallel_tasks, args, device, distributed=False, rank=0,
          world_size=1):
  torch.manual_seed(0)  # same init on every rank
  dev_type = "gpu" if device.type == "cuda" else "cpu"
  base = _StatePolicy(args.state_dim, args.action_dim, args.hidden,
                      device_type=dev_type)
  model = _BenchMAML(base_model=base, device_type=dev_type,
                     compute_dtype="float32",
                     num_inner_loop_steps=args.inner_steps,
                     use_second_order=True,
                     parallel_tasks=parallel_tasks)
  _ = model.network
  model.to_device(device)
  features, labels = _meta_batch(args.tasks, args.samples,
                                 args.state_dim, args.action_dim, device,
                                 seed=1234 + rank)
  opt = torch.optim.Adam(model.network.parameters(), lr=1e-3)

  def one_step():
    opt.zero_grad(set_to_none=True)
    ops = model.model_fn(features, labels, run_modes.TRAIN)
    ops.loss.backward()
    if distributed:
      import torch.distributed as dist
      for p_ in model.network.parameters():
        if p_.grad is not None:
          dist.all_reduce(p_.grad)
          p_.grad.div_(world_size)
    opt.step()
    return ops.loss

  for _ in range(args.warmup):
    one_step()
  if device.type == "cuda":
    torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(args.steps):
    loss = one_step()
  if device.type == "cuda":
    torch.cuda.synchronize()
  dt = (time.perf_counter() - t0) / args.steps
  return {
      "path": "vmap" if parallel_tasks else "loop",
      "ms_per_step": round(dt * 1e3, 3),
      "task_steps_per_sec": round(args.tasks / dt, 1),
      "loss": round(float(loss), 5),
      "tasks": args.tasks, "samples": args.samples,
      "inner_steps": args.inner_steps, "second_order": True,
      "device": device.type,
  }


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--tasks", type=int, default=16)
  p.add_argument("--samples", type=int, default=40)
  p.add_argument("--inner-steps", type=int, default=2)
  p.add_argument("--state-dim", type=int, default=32)
  p.add_argument("--action-dim", type=int, default=7)
  p.add_argument("--hidden", type=int, default=256)
  p.add_argument("--steps", type=int, default=30)
  p.add_argument("--warmup", type=int, default=10)
  p.add_argument("--mode", choices=["both", "loop", "vmap"],
                 default="both")
  args = p.parse_args()
  import dist_bench
  distributed, rank, world_size, device = dist_bench.init()
  results = []
  if args.mode in ("both", "loop"):
    results.append(bench(False, args, device, distributed, rank,
                         world_size))
  if args.mode in ("both", "vmap"):
    results.append(bench(True, args, device, distributed, rank,
                         world_size))
  if rank == 0:
    for r in results:
      r["n_gpus"] = world_size
      r["parallelism"] = "dp%d" % world_size
      r["task_steps_per_sec"] = round(
          r["task_steps_per_sec"] * world_size, 1)
      print(json.dumps(r))
    if len(results) == 2:
      print(json.dumps({
          "speedup_vmap_over_loop": round(
              results[0]["ms_per_step"] / results[1]["ms_per_step"],
              2)}))
  dist_bench.finalize(distributed)


if __name__ == "__main__":
  main()
