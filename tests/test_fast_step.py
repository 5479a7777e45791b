"""FastStepEngine + Trainer fast-path tests (CPU; gloo for multi-rank).

The engine is the framework home of the graphed/flat-grad train step
(parallel/fast_step.py); GPU behavior (hipGraph capture, RCCL) is
covered by tests/test_models_gpu.py — here we prove the step semantics
and the distributed demotion/eager path are numerically correct.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from tensor2robot_amd.models import optimizers
from tensor2robot_amd.parallel import fast_step
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.train import train_eval
from tensor2robot_amd.utils import mocks
from tensor2robot_amd.utils import modes as run_modes

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_net(seed=3):
  torch.manual_seed(seed)
  return torch.nn.Sequential(
      torch.nn.Linear(4, 8), torch.nn.ReLU(), torch.nn.Linear(8, 1))


def _struct(x, y):
  f = tsu.TensorSpecStruct()
  f["x"] = x
  l = tsu.TensorSpecStruct()
  l["y"] = y
  return f, l


def _mse_loss_fn(net):
  def loss_fn(features, labels):
    return torch.nn.functional.mse_loss(net(features["x"]), labels["y"])
  return loss_fn


def test_engine_eager_matches_plain_loop():
  g = torch.Generator().manual_seed(0)
  batches = [(torch.randn(6, 4, generator=g),
              torch.randn(6, 1, generator=g)) for _ in range(4)]

  net_a = _tiny_net()
  opt_a = torch.optim.SGD(net_a.parameters(), lr=0.05)
  for x, y in batches:
    opt_a.zero_grad()
    torch.nn.functional.mse_loss(net_a(x), y).backward()
    opt_a.step()

  net_b = _tiny_net()

  class _Opt:  # ScheduledOptimizer-shaped wrapper
    def __init__(self, params):
      self.opt = torch.optim.SGD(params, lr=0.05)

    def zero_grad(self, set_to_none=True):
      self.opt.zero_grad(set_to_none=set_to_none)

    def step(self, global_step):
      self.opt.step()

  opt_b = _Opt(net_b.parameters())
  engine = fast_step.FastStepEngine(net_b, opt_b, ema=None,
                                    device=torch.device("cpu"),
                                    use_graph=False)
  f0, l0 = _struct(*batches[0])
  engine.build(_mse_loss_fn(net_b), f0, l0)
  for x, y in batches:
    f, l = _struct(x, y)
    engine.step(f, l)

  for pa, pb in zip(net_a.parameters(), net_b.parameters()):
    torch.testing.assert_close(pa, pb)


def _dist_worker(rank, world, init_file, out_dir):
  import torch.distributed as dist
  dist.init_process_group(backend="gloo",
                          init_method=f"file://{init_file}",
                          rank=rank, world_size=world)
  try:
    net = _tiny_net(seed=100 + rank)  # different init; engine broadcasts
    opt = torch.optim.SGD(net.parameters(), lr=0.05)

    class _Opt:
      def zero_grad(self, set_to_none=True):
        opt.zero_grad(set_to_none=set_to_none)

      def step(self, global_step):
        opt.step()

    engine = fast_step.FastStepEngine(net, _Opt(), ema=None,
                                      device=torch.device("cpu"),
                                      use_graph=False)
    g = torch.Generator().manual_seed(200 + rank)
    x = torch.randn(5, 4, generator=g)
    y = torch.randn(5, 1, generator=g)
    f, l = _struct(x, y)
    engine.build(_mse_loss_fn(net), f, l)
    for _ in range(3):
      engine.step(f, l)
    torch.save({k: v.clone() for k, v in net.state_dict().items()},
               os.path.join(out_dir, f"rank{rank}.pt"))
  finally:
    dist.destroy_process_group()


def test_engine_dist_gloo_matches_global_batch(tmp_path):
  world = 2
  ctx = mp.get_context("spawn")
  procs = [ctx.Process(target=_dist_worker,
                       args=(r, world, str(tmp_path / "init"),
                             str(tmp_path)))
           for r in range(world)]
  for p in procs:
    p.start()
  for p in procs:
    p.join(timeout=120)
    assert p.exitcode == 0
  s0 = torch.load(tmp_path / "rank0.pt", weights_only=False)
  s1 = torch.load(tmp_path / "rank1.pt", weights_only=False)
  for k in s0:
    assert torch.equal(s0[k], s1[k]), k

  # Single-process global-batch reference: rank-0 init (broadcast),
  # loss = mean of per-rank losses (what all_reduce SUM / world does).
  net = _tiny_net(seed=100)
  opt = torch.optim.SGD(net.parameters(), lr=0.05)
  data = []
  for rank in range(world):
    g = torch.Generator().manual_seed(200 + rank)
    data.append((torch.randn(5, 4, generator=g),
                 torch.randn(5, 1, generator=g)))
  for _ in range(3):
    opt.zero_grad()
    loss = sum(torch.nn.functional.mse_loss(net(x), y)
               for x, y in data) / world
    loss.backward()
    opt.step()
  for k, v in net.state_dict().items():
    torch.testing.assert_close(s0[k], v, rtol=1e-5, atol=1e-6)


def _eval_worker(rank, world, init_file, out_dir):
  import torch.distributed as dist
  dist.init_process_group(backend="gloo",
                          init_method=f"file://{init_file}",
                          rank=rank, world_size=world)
  try:
    torch.manual_seed(0)  # identical weights on every rank + reference
    model = mocks.MockT2RModel(
        device_type="cpu",
        create_optimizer_fn=lambda: optimizers.create_adam_optimizer(
            learning_rate=1e-2))
    trainer = train_eval.Trainer(model, model_dir="")
    gen = mocks.MockInputGenerator(batch_size=4, seed=40 + rank)
    gen.set_specification_from_model(model, run_modes.EVAL)
    input_fn = gen.create_dataset_input_fn(run_modes.EVAL)
    metrics = trainer.evaluate(input_fn, eval_steps=3,
                               distributed_eval=True)
    torch.save(metrics, os.path.join(out_dir, f"metrics{rank}.pt"))
  finally:
    dist.destroy_process_group()


def test_distributed_eval_reduces_metrics(tmp_path):
  """Sharded eval across 2 ranks == pooled single-process eval
  (Trainer.evaluate cross-rank metric reduction)."""
  world = 2
  ctx = mp.get_context("spawn")
  procs = [ctx.Process(target=_eval_worker,
                       args=(r, world, str(tmp_path / "init"),
                             str(tmp_path)))
           for r in range(world)]
  for p in procs:
    p.start()
  for p in procs:
    p.join(timeout=180)
    assert p.exitcode == 0
  m0 = torch.load(tmp_path / "metrics0.pt", weights_only=False)
  m1 = torch.load(tmp_path / "metrics1.pt", weights_only=False)
  assert m0.keys() == m1.keys()
  for k in m0:
    assert m0[k] == pytest.approx(m1[k]), k

  # Pooled reference: evaluate both shards in ONE process with a fresh
  # (deterministic-init) model identical to the workers' models.
  torch.manual_seed(0)
  model = mocks.MockT2RModel(
      device_type="cpu",
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(
          learning_rate=1e-2))
  trainer = train_eval.Trainer(model, model_dir="")
  totals, count = {}, 0
  for rank in range(world):
    gen = mocks.MockInputGenerator(batch_size=4, seed=40 + rank)
    gen.set_specification_from_model(model, run_modes.EVAL)
    input_fn = gen.create_dataset_input_fn(run_modes.EVAL)
    m = trainer.evaluate(input_fn, eval_steps=3)
    for k, v in m.items():
      if k == "global_step":
        continue
      totals[k] = totals.get(k, 0.0) + v
    count += 1
  pooled = {k: v / count for k, v in totals.items()}
  for k, v in pooled.items():
    assert m0[k] == pytest.approx(v, abs=1e-6), k


def test_trainer_deferred_preprocess_matches_pipeline(tmp_path):
  """Deferring the preprocessor to the trainer (the GPU fast path's
  structure) must train identically to in-pipeline preprocessing."""
  import functools

  def make(seed):
    torch.manual_seed(seed)
    model = mocks.MockT2RModel(
        device_type="cpu",
        create_optimizer_fn=lambda: optimizers.create_adam_optimizer(
            learning_rate=1e-2))
    gen = mocks.MockInputGenerator(batch_size=8, seed=7)
    gen.set_specification_from_model(model, run_modes.TRAIN)
    return model, gen

  # In-pipeline preprocessing (CPU default path).
  model_a, gen_a = make(0)
  trainer_a = train_eval.Trainer(model_a, model_dir="")
  trainer_a.train(gen_a.create_dataset_input_fn(run_modes.TRAIN),
                  max_steps=5)

  # Deferred: generator yields raw batches, trainer applies the fn.
  model_b, gen_b = make(0)
  fn = gen_b.defer_preprocessing()
  assert fn is not None
  trainer_b = train_eval.Trainer(model_b, model_dir="")
  trainer_b.train(gen_b.create_dataset_input_fn(run_modes.TRAIN),
                  max_steps=5, preprocess_fn=fn)

  for pa, pb in zip(model_a.network.parameters(),
                    model_b.network.parameters()):
    torch.testing.assert_close(pa, pb)
