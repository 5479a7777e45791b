"""Shared distributed plumbing for the family benches (DP over RCCL).

Gives tools/bench_bcz.py, bench_grasp2vec.py and bench_maml.py the same
launch contract as bench.py (BASELINE configs #3-#5 at DP=N):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/bench_<family>.py ...

One rank per GPU; rank state comes from the torchrun env.  On CPU the
backend is gloo (the ws=2 smoke tests in tests/test_family_dist.py).
"""

import os

import torch
import torch.distributed as dist


def init():
  """Returns (distributed, rank, world_size, device)."""
  world_size = int(os.environ.get("WORLD_SIZE", "1"))
  rank = int(os.environ.get("RANK", "0"))
  local_rank = int(os.environ.get("LOCAL_RANK", "0"))
  distributed = world_size > 1 or bool(os.environ.get("T2R_FORCE_DIST"))
  use_cuda = torch.cuda.is_available()
  if distributed:
    dist.init_process_group(backend="nccl" if use_cuda else "gloo")
  if use_cuda:
    torch.cuda.set_device(torch.device(f"cuda:{local_rank}"))
    device = torch.device(f"cuda:{local_rank}")
  else:
    device = torch.device("cpu")
  return distributed, rank, world_size, device


def barrier_sync(distributed: bool):
  if distributed:
    dist.barrier()
  if torch.cuda.is_available():
    torch.cuda.synchronize()


def max_over_ranks(elapsed: float, distributed: bool, device) -> float:
  if not distributed:
    return elapsed
  t = torch.tensor([elapsed], dtype=torch.float64,
                   device=device if device.type == "cuda" else "cpu")
  dist.all_reduce(t, op=dist.ReduceOp.MAX)
  return float(t.item())


def finalize(distributed: bool):
  if distributed:
    dist.destroy_process_group()
