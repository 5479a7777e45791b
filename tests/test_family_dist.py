"""ws=2 torchrun smokes for the family bench entry points (CPU/gloo).

BASELINE configs #3-#5 are DP=8 jobs; these run each family bench
under the driver's exact multi-rank launch shape (torch.distributed.run
--nnodes=1) at world_size 2 on gloo, verifying init, per-rank data,
gradient all-reduce and the rank-0 whole-job report.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _torchrun(script, extra, port):
  env = dict(os.environ)
  env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
  cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), script] + extra
  out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=600)
  assert out.returncode == 0, out.stderr[-4000:]
  lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
  assert lines, out.stdout[-2000:]
  return [json.loads(l) for l in lines]


@pytest.mark.slow
def test_bench_bcz_torchrun_ws2_cpu():
  results = _torchrun("tools/bench_bcz.py",
                      ["--steps", "1", "--warmup", "0",
                       "--batch-size", "2"], 29521)
  r = results[0]
  assert r["n_gpus"] == 2 and r["parallelism"] == "dp2"
  assert r["value"] > 0


@pytest.mark.slow
def test_bench_grasp2vec_torchrun_ws2_cpu():
  results = _torchrun("tools/bench_grasp2vec.py",
                      ["--steps", "1", "--warmup", "0",
                       "--batch-size", "2", "--image-size", "48",
                       "--resnet-size", "18"], 29522)
  r = results[0]
  assert r["n_gpus"] == 2 and r["parallelism"] == "dp2"
  assert r["value"] > 0


@pytest.mark.slow
def test_bench_maml_torchrun_ws2_cpu():
  results = _torchrun("tools/bench_maml.py",
                      ["--steps", "2", "--warmup", "1", "--tasks", "2",
                       "--samples", "4", "--hidden", "32"], 29523)
  assert len(results) == 3  # loop, vmap, speedup
  for r in results[:2]:
    assert r["n_gpus"] == 2 and r["parallelism"] == "dp2"
