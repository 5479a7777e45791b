"""Multi-process (gloo, world_size=2) tests of the DP gradient engine.

The round-end scaling bench launches one rank per GPU over RCCL; this suite
proves the same code path (DataParallelEngine bucketing + all-reduce,
bench.py's step loop) is numerically correct on CPU with gloo so the RCCL
run is correct by construction. Mirrors the reference's single-process
"distributed" test philosophy but actually spins up ranks.
"""

import json
import os
import subprocess
import sys
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make_model(seed: int) -> torch.nn.Module:
  g = torch.Generator().manual_seed(seed)
  torch.manual_seed(seed)
  return torch.nn.Sequential(
      torch.nn.Linear(8, 16), torch.nn.ReLU(),
      torch.nn.Linear(16, 4), torch.nn.ReLU(),
      torch.nn.Linear(4, 1))


def _rank_data(rank: int, n: int = 6):
  g = torch.Generator().manual_seed(100 + rank)
  x = torch.randn(n, 8, generator=g)
  y = torch.randn(n, 1, generator=g)
  return x, y


def _worker(rank, world_size, init_file, out_dir, bucket_cap_mb):
  import torch.distributed as dist
  from tensor2robot_amd.parallel import ddp
  dist.init_process_group(
      backend="gloo", init_method=f"file://{init_file}",
      rank=rank, world_size=world_size)
  try:
    # Deliberately different init per rank: the engine must broadcast
    # rank 0's weights so replicas start identical.
    model = _make_model(seed=1000 + rank)
    engine = ddp.DataParallelEngine(model, bucket_cap_mb=bucket_cap_mb)
    x, y = _rank_data(rank)
    for _ in range(3):
      loss = torch.nn.functional.mse_loss(model(x), y)
      for p in model.parameters():
        p.grad = None
      engine.backward(loss)
      with torch.no_grad():
        for p in model.parameters():
          p -= 0.05 * p.grad
    state = {k: v.clone() for k, v in model.state_dict().items()}
    grads = [p.grad.clone() for p in model.parameters()]
    torch.save({"state": state, "grads": grads},
               os.path.join(out_dir, f"rank{rank}.pt"))
  finally:
    dist.destroy_process_group()


def _single_process_reference():
  """Same 3 steps on the concatenated data of both ranks (DP equivalence)."""
  model = _make_model(seed=1000)  # rank 0 init is the broadcast one
  xs, ys = zip(_rank_data(0), _rank_data(1))
  for _ in range(3):
    for p in model.parameters():
      p.grad = None
    # Average of per-rank losses == what all-reduce(SUM)/world computes.
    loss = 0.5 * (
        torch.nn.functional.mse_loss(model(xs[0]), ys[0]) +
        torch.nn.functional.mse_loss(model(xs[1]), ys[1]))
    loss.backward()
    with torch.no_grad():
      for p in model.parameters():
        p -= 0.05 * p.grad
  return model


@pytest.mark.parametrize("bucket_cap_mb", [128.0, 0.0001])
def test_dp_engine_matches_single_process(tmp_path, bucket_cap_mb):
  world = 2
  init_file = str(tmp_path / "init")
  ctx = mp.get_context("spawn")
  procs = []
  for r in range(world):
    p = ctx.Process(target=_worker,
                    args=(r, world, init_file, str(tmp_path), bucket_cap_mb))
    p.start()
    procs.append(p)
  for p in procs:
    p.join(timeout=120)
    assert p.exitcode == 0

  r0 = torch.load(tmp_path / "rank0.pt", weights_only=False)
  r1 = torch.load(tmp_path / "rank1.pt", weights_only=False)
  # Replicas stay bit-identical in weights and reduced grads.
  for k in r0["state"]:
    assert torch.equal(r0["state"][k], r1["state"][k]), k
  for g0, g1 in zip(r0["grads"], r1["grads"]):
    assert torch.equal(g0, g1)
  # And the result matches the single-process global-batch reference.
  ref = _single_process_reference()
  for k, v in ref.state_dict().items():
    torch.testing.assert_close(r0["state"][k], v, rtol=1e-5, atol=1e-6)


@pytest.mark.slow
def test_bench_under_torchrun_cpu(tmp_path):
  """The driver's exact multi-rank launch shape, on CPU/gloo."""
  env = dict(os.environ)
  env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
  cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29511", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch-size", "2"]
  out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=600)
  assert out.returncode == 0, out.stderr[-4000:]
  line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
  result = json.loads(line)
  assert result["n_gpus"] == 2
  assert result["value"] > 0
  assert result["config"]["parallelism"] == "dp2"


def _flatview_worker(rank, world_size, init_file, out_dir):
  """bench.py's graphed-dist gradient flow, minus the graphs: p.grad
  pre-assigned as strided views into one flat comm buffer (channels_last
  strides for 4D params), backward ACCUMULATES into it, one all-reduce
  of the buffer syncs every grad."""
  import torch.distributed as dist
  dist.init_process_group(
      backend="gloo", init_method=f"file://{init_file}",
      rank=rank, world_size=world_size)
  try:
    torch.manual_seed(7)   # identical init on both ranks
    model = torch.nn.Sequential(
        torch.nn.Conv2d(4, 8, 3, padding=1, bias=False),
        torch.nn.ReLU(), torch.nn.Flatten(),
        torch.nn.Linear(8 * 4 * 4, 2))
    model[0].to(memory_format=torch.channels_last)
    params = [p for p in model.parameters() if p.requires_grad]
    flat = torch.zeros(sum(p.numel() for p in params))
    off = 0
    for p in params:
      sl = flat[off:off + p.numel()]
      if p.dim() == 4 and p.is_contiguous(
          memory_format=torch.channels_last):
        no, c, h, w = p.shape
        g = sl.view(no, h, w, c).permute(0, 3, 1, 2)
      else:
        g = sl.view(p.shape)
      p.grad = g
      off += p.numel()

    gdata = torch.Generator().manual_seed(500 + rank)
    x = torch.randn(3, 4, 4, 4, generator=gdata)
    y = torch.randn(3, 2, generator=gdata)
    for _ in range(2):
      flat.zero_()
      loss = torch.nn.functional.mse_loss(model(x), y)
      loss.backward()
      # grads must still alias the comm buffer (the accumulate-into-
      # view mechanism bench.py asserts on).
      base = flat.data_ptr()
      end = base + flat.numel() * flat.element_size()
      for p in params:
        assert base <= p.grad.data_ptr() < end
      dist.all_reduce(flat)
      flat.div_(world_size)
      with torch.no_grad():
        for p in params:
          p -= 0.1 * p.grad
    torch.save({k: v.clone() for k, v in model.state_dict().items()},
               os.path.join(out_dir, f"fv_rank{rank}.pt"))
  finally:
    dist.destroy_process_group()


def test_flat_grad_view_sync_matches_reference(tmp_path):
  world = 2
  init_file = str(tmp_path / "init_fv")
  ctx = mp.get_context("spawn")
  procs = []
  for r in range(world):
    p = ctx.Process(target=_flatview_worker,
                    args=(r, world, init_file, str(tmp_path)))
    p.start()
    procs.append(p)
  for p in procs:
    p.join(timeout=120)
    assert p.exitcode == 0
  r0 = torch.load(tmp_path / "fv_rank0.pt", weights_only=False)
  r1 = torch.load(tmp_path / "fv_rank1.pt", weights_only=False)
  for k in r0:
    assert torch.equal(r0[k], r1[k]), k

  # Single-process global-batch reference.
  torch.manual_seed(7)
  ref = torch.nn.Sequential(
      torch.nn.Conv2d(4, 8, 3, padding=1, bias=False),
      torch.nn.ReLU(), torch.nn.Flatten(),
      torch.nn.Linear(8 * 4 * 4, 2))
  ref[0].to(memory_format=torch.channels_last)
  xs, ys = [], []
  for r in range(world):
    g = torch.Generator().manual_seed(500 + r)
    xs.append(torch.randn(3, 4, 4, 4, generator=g))
    ys.append(torch.randn(3, 2, generator=g))
  for _ in range(2):
    for p in ref.parameters():
      p.grad = None
    loss = 0.5 * sum(
        torch.nn.functional.mse_loss(ref(xs[r]), ys[r])
        for r in range(world))
    loss.backward()
    with torch.no_grad():
      for p in ref.parameters():
        p -= 0.1 * p.grad
  for k, v in ref.state_dict().items():
    torch.testing.assert_close(r0[k], v, rtol=1e-5, atol=1e-6)
