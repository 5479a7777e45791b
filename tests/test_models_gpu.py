"""GPU tests for the layer library and model families (bf16, fused BN).

Each numerics test compares the HIP/bf16 path against a plain-torch fp32
reference of the same op/net.
"""

import copy

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
def test_resnet18_gpu_bf16_matches_cpu_fp32():
  from tensor2robot_amd.layers import resnet
  torch.manual_seed(0)
  net_cpu = resnet.ResNet(resnet_size=18, num_classes=8).eval()
  net_gpu = copy.deepcopy(net_cpu).cuda().to(
      memory_format=torch.channels_last).eval()
  x = torch.randn(4, 3, 64, 64)
  with torch.no_grad():
    out_cpu, _ = net_cpu(x)
    with torch.autocast("cuda", dtype=torch.bfloat16):
      out_gpu, _ = net_gpu(x.cuda().contiguous(
          memory_format=torch.channels_last))
  assert torch.allclose(out_cpu, out_gpu.float().cpu(), atol=0.1,
                        rtol=0.1), \
      float((out_cpu - out_gpu.float().cpu()).abs().max())


@requires_gpu
def test_resnet_train_step_fused_bn_runs():
  from tensor2robot_amd.layers import resnet
  torch.manual_seed(0)
  net = resnet.ResNet(resnet_size=18, num_classes=4).cuda().to(
      memory_format=torch.channels_last).train()
  opt = torch.optim.SGD(net.parameters(), lr=1e-2)
  x = torch.randn(8, 3, 64, 64, device="cuda").contiguous(
      memory_format=torch.channels_last)
  with torch.autocast("cuda", dtype=torch.bfloat16):
    out, _ = net(x)
    loss = out.float().pow(2).mean()
  loss.backward()
  opt.step()
  for p in net.parameters():
    assert torch.isfinite(p).all()
  # BN running stats moved.
  bn = net.block_layers[0].blocks[0].bn1
  assert not torch.allclose(bn.running_mean,
                            torch.zeros_like(bn.running_mean))


@requires_gpu
def test_spatial_softmax_gpu_matches_cpu():
  from tensor2robot_amd.layers import spatial_softmax
  torch.manual_seed(0)
  feat = torch.randn(3, 8, 14, 14)
  p_cpu, _ = spatial_softmax.SpatialSoftmax()(feat)
  p_gpu, _ = spatial_softmax.SpatialSoftmax()(feat.cuda())
  assert torch.allclose(p_cpu, p_gpu.cpu(), atol=1e-5, rtol=1e-4)


@requires_gpu
def test_bcz_train_step_gpu_bf16():
  import functools
  from tensor2robot_amd.models import optimizers
  from tensor2robot_amd.research.bcz import model as bcz_model
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  from tensor2robot_amd.utils import modes as run_modes
  torch.manual_seed(0)
  model = bcz_model.BCZModel(
      image_size=(100, 100), input_size=(512, 640),
      preprocessor_cls=functools.partial(
          bcz_model.BCZPreprocessor, image_size=(100, 100),
          input_size=(512, 640), crop_size=(472, 472),
          mock_subtask=True),
      create_optimizer_fn=lambda: optimizers.create_adam_optimizer(1e-4),
      device_type="gpu", compute_dtype="bfloat16")
  model.to_device(torch.device("cuda:0"))
  model.network.to(memory_format=torch.channels_last)
  g = torch.Generator().manual_seed(0)
  features = tsu.TensorSpecStruct()
  features["image"] = torch.randint(0, 256, (4, 512, 640, 3),
                                    dtype=torch.uint8,
                                    generator=g).cuda()
  features["subtask_id"] = torch.zeros(4, 1, dtype=torch.int64).cuda()
  for name, size, _, _ in model._action_components:
    features["present/" + name] = torch.rand(4, size,
                                             generator=g).cuda()
  labels = tsu.TensorSpecStruct()
  labels["future/xyz_residual"] = torch.randn(4, 1, 3,
                                              generator=g).cuda()
  labels["future/quaternion"] = torch.randn(4, 1, 4, generator=g).cuda()
  labels["future/target_close"] = (torch.rand(4, 1, 1, generator=g)
                                   > 0.5).float().cuda()
  with torch.autocast("cuda", dtype=torch.bfloat16):
    f, l = model.preprocessor.preprocess(features, labels,
                                         run_modes.TRAIN)
    ops = model.model_fn(f, l, run_modes.TRAIN)
  ops.loss.backward()
  torch.cuda.synchronize()
  assert torch.isfinite(ops.loss)


@requires_gpu
def test_grasp2vec_step_gpu():
  from tensor2robot_amd.research.grasp2vec import grasp2vec_model
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  from tensor2robot_amd.utils import modes as run_modes
  torch.manual_seed(0)
  model = grasp2vec_model.Grasp2VecModel(
      scene_size=(128, 128), goal_size=(128, 128), resnet_size=18,
      device_type="gpu", compute_dtype="bfloat16")
  model.to_device(torch.device("cuda:0"))
  model.network.to(memory_format=torch.channels_last)
  features = tsu.TensorSpecStruct()
  for key in ("pregrasp_image", "postgrasp_image", "goal_image"):
    features[key] = torch.rand(4, 128, 128, 3, device="cuda")
  with torch.autocast("cuda", dtype=torch.bfloat16):
    ops = model.model_fn(features, None, run_modes.TRAIN)
  ops.loss.backward()
  torch.cuda.synchronize()
  assert torch.isfinite(ops.loss)


@requires_gpu
def test_maml_inner_loop_gpu():
  from tensor2robot_amd.meta_learning import maml_inner_loop
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  class _Base:
    def __init__(self):
      torch.manual_seed(0)
      self.network = torch.nn.Linear(4, 2).cuda()

    def inference_network_fn(self, features, labels, mode, params=None):
      return {"prediction": self.network(features["x"])}

    def model_train_fn(self, features, labels, inference_outputs, mode,
                       params=None):
      return torch.nn.functional.mse_loss(
          inference_outputs["prediction"], labels["y"])

  base = _Base()
  loop = maml_inner_loop.MAMLInnerLoopGradientDescent(learning_rate=0.05)
  f = tsu.TensorSpecStruct()
  f["x"] = torch.randn(8, 4, device="cuda")
  l = tsu.TensorSpecStruct()
  l["y"] = torch.randn(8, 2, device="cuda")
  outputs, _, inner_losses = loop.inner_loop(
      [(f, l), (f, l), (f, l)], base.inference_network_fn,
      base.model_train_fn, base.network, mode="train")
  assert float(inner_losses[-1]) < float(inner_losses[0])
  outputs[1]["prediction"].pow(2).mean().backward()
  assert base.network.weight.grad is not None


@requires_gpu
def test_graphed_step_trains_grasping44():
  """hipGraph replay must actually update weights (loss decreases)."""
  from tensor2robot_amd.parallel import graph_step
  from tensor2robot_amd.models import optimizers as optimizers_mod
  from tensor2robot_amd.research.qtopt import t2r_models, networks
  torch.manual_seed(0)
  device = torch.device("cuda:0")
  network = networks.Grasping44(action_dim=t2r_models.ACTION_DIM).to(
      device).to(memory_format=torch.channels_last).train()
  opt_factory = optimizers_mod.create_momentum_optimizer(
      learning_rate=1e-3, momentum=0.9)
  optimizer = opt_factory(network.parameters())
  bs = 8
  image = torch.rand(bs, 3, 472, 472, device=device).to(
      torch.bfloat16).contiguous(memory_format=torch.channels_last)
  action = torch.rand(bs, t2r_models.ACTION_DIM, device=device).to(
      torch.bfloat16)
  # Learnable rule: success iff mean pixel > 0.5 region — use fixed labels.
  labels = (torch.rand(bs, device=device) > 0.5).float()
  autocast = torch.autocast("cuda", dtype=torch.bfloat16)

  def body():
    # set_to_none=True inside capture: backward writes stable
    # graph-pool grad buffers (the bench.py graph-body contract).
    optimizer.zero_grad(set_to_none=True)
    with autocast:
      logit = network(image, action)
    loss = torch.nn.functional.binary_cross_entropy_with_logits(
        logit.float(), labels.reshape(logit.shape))
    loss.backward()
    optimizer.step(0)
    return loss

  for _ in range(3):
    body()
  torch.cuda.synchronize()
  graphed = graph_step.GraphedTrainStep(body)
  first = float(graphed.replay())
  params_before = [p.detach().clone() for p in network.parameters()]
  for _ in range(30):
    graphed.replay()
  last = float(graphed.replay())
  torch.cuda.synchronize()
  assert last < first, (first, last)  # memorizes fixed labels
  changed = sum(
      0 if torch.equal(p.detach(), q) else 1
      for p, q in zip(network.parameters(), params_before))
  assert changed > 0


def _qtopt_pool(batch_size, device, n_batches=2, seed=0):
  import itertools
  from tensor2robot_amd.research.qtopt import t2r_models
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  g = torch.Generator().manual_seed(seed)
  pool = []
  for _ in range(n_batches):
    f = tsu.TensorSpecStruct()
    f["state/image"] = torch.randint(
        0, 256, (batch_size, t2r_models.RAW_HEIGHT, t2r_models.RAW_WIDTH,
                 3), generator=g, dtype=torch.uint8).to(device)
    off = 0
    act = torch.rand(batch_size, t2r_models.ACTION_DIM, generator=g)
    for name, size in t2r_models.ACTION_COMPONENTS:
      f["action/" + name] = act[:, off:off + size].to(device)
      off += size
    l = tsu.TensorSpecStruct()
    l["grasp_success"] = (torch.rand(batch_size, 1, generator=g) >
                          0.5).float().to(device)
    pool.append((f, l))
  return itertools.cycle(pool)


@requires_gpu
def test_trainer_fast_step_qtopt_graphed():
  """The Trainer's default GPU path must capture hipGraphs and train
  the flagship model (the fast step is the product)."""
  import functools
  from tensor2robot_amd.research.qtopt import t2r_models
  from tensor2robot_amd.train import train_eval
  from tensor2robot_amd.utils import modes as run_modes
  torch.manual_seed(0)
  model = t2r_models.GraspingModel(device_type="gpu",
                                   compute_dtype="bfloat16",
                                   use_avg_model_params=True)
  trainer = train_eval.Trainer(model, model_dir="")
  pool = _qtopt_pool(8, trainer.device)
  preprocess_fn = functools.partial(model.preprocessor.preprocess,
                                    mode=run_modes.TRAIN)
  trainer.train(lambda: pool, max_steps=10, preprocess_fn=preprocess_fn)
  assert trainer._fast_engine is not None
  assert trainer._fast_engine.is_graphed, "hipGraph capture did not run"
  ops = trainer._captured_ops
  assert ops is not None and torch.isfinite(ops.loss)


@requires_gpu
def test_trainer_forced_dist_ws1_rccl_smoke(tmp_path):
  """world_size-1 RCCL: init + broadcast + flat-buffer all_reduce + the
  two-graph distributed step run end to end on one GPU (VERDICT item 7:
  exercise the dist path on every round's GPU tier)."""
  import functools
  import torch.distributed as dist
  from tensor2robot_amd.research.qtopt import t2r_models
  from tensor2robot_amd.train import train_eval
  from tensor2robot_amd.utils import modes as run_modes
  assert not dist.is_initialized()
  dist.init_process_group(
      backend="nccl", init_method=f"file://{tmp_path}/init",
      rank=0, world_size=1)
  try:
    model = t2r_models.GraspingModel(device_type="gpu",
                                     compute_dtype="bfloat16",
                                     use_avg_model_params=True)
    trainer = train_eval.Trainer(model, model_dir="")
    pool = _qtopt_pool(4, trainer.device)
    preprocess_fn = functools.partial(model.preprocessor.preprocess,
                                      mode=run_modes.TRAIN)
    trainer.train(lambda: pool, max_steps=6, preprocess_fn=preprocess_fn)
    engine = trainer._fast_engine
    assert engine.distributed
    assert engine.is_graphed
    assert engine.opt_graphed is not None
    assert engine._flat is not None  # flat comm buffer exists
    # grads alias the comm buffer (accumulate-into-view mechanism).
    engine._check_grad_aliasing()
  finally:
    dist.destroy_process_group()


@requires_gpu
def test_engine_graphed_matches_eager_steps():
  """Graphed and eager engine runs produce the same weights on the
  same fixed batch (eager runs 6 extra steps = the graphed build's 3
  settle + 3 capture-warmup updates)."""
  from tensor2robot_amd.parallel import fast_step
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  def make():
    torch.manual_seed(3)
    net = torch.nn.Sequential(
        torch.nn.Conv2d(8, 16, 3, padding=1, bias=False),
        torch.nn.ReLU(), torch.nn.Flatten(),
        torch.nn.Linear(16 * 8 * 8, 1)).to("cuda").to(
            memory_format=torch.channels_last)
    opt = torch.optim.SGD(net.parameters(), lr=0.01)

    class _Opt:
      def zero_grad(self, set_to_none=True):
        opt.zero_grad(set_to_none=set_to_none)

      def step(self, global_step):
        opt.step()
    return net, _Opt()

  g = torch.Generator().manual_seed(0)
  x = torch.randn(4, 8, 8, 8, generator=g).to("cuda").to(
      memory_format=torch.channels_last)
  y = torch.randn(4, 1, generator=g).to("cuda")
  f = tsu.TensorSpecStruct(); f["x"] = x
  l = tsu.TensorSpecStruct(); l["y"] = y

  def loss_fn_for(net):
    def loss_fn(features, labels):
      return torch.nn.functional.mse_loss(
          net(features["x"]).float(), labels["y"])
    return loss_fn

  K = 5
  net_g, opt_g = make()
  eng_g = fast_step.FastStepEngine(net_g, opt_g, use_graph=True,
                                   autocast_dtype=torch.bfloat16)
  eng_g.build(loss_fn_for(net_g), f, l)
  assert eng_g.is_graphed
  for _ in range(K):
    eng_g.step(f, l)
  torch.cuda.synchronize()

  net_e, opt_e = make()
  eng_e = fast_step.FastStepEngine(net_e, opt_e, use_graph=False,
                                   autocast_dtype=torch.bfloat16)
  eng_e.build(loss_fn_for(net_e), f, l)
  for _ in range(6 + K):  # 3 settle + 3 graph-warmup + K
    eng_e.step(f, l)
  torch.cuda.synchronize()

  for pg, pe in zip(net_g.parameters(), net_e.parameters()):
    torch.testing.assert_close(pg, pe, rtol=3e-2, atol=1e-4)


@requires_gpu
def test_train_eval_model_step_parity_with_device_pool():
  """train_eval_model()'s full path (pipeline + H2D + deferred GPU
  preprocess + graphed step) must hold ms/step close to the device-
  resident-pool loop bench.py times (VERDICT item 3 done criterion)."""
  import functools
  import time
  from tensor2robot_amd.data import input_generators
  from tensor2robot_amd.research.qtopt import t2r_models
  from tensor2robot_amd.specs import tensorspec_utils as tsu
  from tensor2robot_amd.train import train_eval
  from tensor2robot_amd.utils import modes as run_modes

  bs = 32  # the flagship batch size: its conv shapes are pinned in the
  # packaged MIOpen DB.  (At off-DB batch sizes MIOpen's find can park
  # on a slow wrw winner for minutes — tracked separately.)

  class _PoolGenerator(input_generators.AbstractInputGenerator):
    """Cycles pregenerated CPU batches (no per-step numpy RNG cost)."""

    def _iterate(self, mode):
      g = torch.Generator().manual_seed(9)
      batches = []
      for _ in range(3):
        f = tsu.TensorSpecStruct()
        f["state/image"] = torch.randint(
            0, 256, (bs, t2r_models.RAW_HEIGHT, t2r_models.RAW_WIDTH, 3),
            generator=g, dtype=torch.uint8)
        off = 0
        act = torch.rand(bs, t2r_models.ACTION_DIM, generator=g)
        for name, size in t2r_models.ACTION_COMPONENTS:
          f["action/" + name] = act[:, off:off + size].clone()
          off += size
        l = tsu.TensorSpecStruct()
        l["grasp_success"] = (torch.rand(bs, 1, generator=g) >
                              0.5).float()
        batches.append((f, l))
      i = 0
      while True:
        yield batches[i % len(batches)]
        i += 1

  def timed_steps(trainer, input_fn, preprocess_fn, n):
    trainer.train(input_fn, trainer.global_step + 25,
                  preprocess_fn=preprocess_fn)  # settle
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    trainer.train(input_fn, trainer.global_step + n,
                  preprocess_fn=preprocess_fn)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000.0

  # Path A: the product — train_eval_model-style pipeline from CPU.
  torch.manual_seed(0)
  model_a = t2r_models.GraspingModel(device_type="gpu",
                                     compute_dtype="bfloat16")
  trainer_a = train_eval.Trainer(model_a, model_dir="")
  gen = _PoolGenerator(batch_size=bs)
  gen.set_specification_from_model(model_a, run_modes.TRAIN)
  deferred = gen.defer_preprocessing()
  input_fn_a = gen.create_dataset_input_fn(run_modes.TRAIN,
                                           pin_memory=True,
                                           h2d_device="cuda")
  ms_pipeline = timed_steps(trainer_a, input_fn_a, deferred, 30)

  # Path B: device-resident pool (bench.py's shape).
  torch.manual_seed(0)
  model_b = t2r_models.GraspingModel(device_type="gpu",
                                     compute_dtype="bfloat16")
  trainer_b = train_eval.Trainer(model_b, model_dir="")
  pool = _qtopt_pool(bs, trainer_b.device, n_batches=3)
  preprocess_fn = functools.partial(model_b.preprocessor.preprocess,
                                    mode=run_modes.TRAIN)
  ms_pool = timed_steps(trainer_b, lambda: pool, preprocess_fn, 30)

  print(f"ms/step pipeline={ms_pipeline:.3f} pool={ms_pool:.3f}")
  # The CPU-source pipeline pays H2D + producer staging on top of the
  # pool path.  Producer-side copy-stream prefetch brought it from
  # ~4.3x to ~3.4x of the pool step; the residual slow-copy effect
  # (in-pipeline H2D runs ~6 GB/s where an isolated under-load copy
  # does 26 GB/s) is an open item (profiles/r2_input_pipeline.md).
  assert ms_pipeline <= ms_pool * 4.0 + 1.0, (ms_pipeline, ms_pool)


@requires_gpu
def test_maml_vmap_parallel_tasks_matches_loop_gpu():
  """Task-parallel (vmap) inner loop == per-task loop on GPU, incl.
  second-order outer grads (BASELINE config #5 workload)."""
  from tensor2robot_amd.meta_learning import maml_inner_loop
  from tensor2robot_amd.specs import tensorspec_utils as tsu

  torch.manual_seed(0)
  net = torch.nn.Sequential(
      torch.nn.Linear(6, 64), torch.nn.ReLU(),
      torch.nn.Linear(64, 64), torch.nn.ReLU(),
      torch.nn.Linear(64, 2)).cuda()

  class _Base:
    network = net

    def inference_network_fn(self, features, labels, mode, params=None):
      return {"prediction": net(features["x"])}

    def model_train_fn(self, features, labels, inference_outputs, mode,
                       params=None):
      return torch.nn.functional.mse_loss(
          inference_outputs["prediction"], labels["y"])

  base = _Base()
  loop = maml_inner_loop.MAMLInnerLoopGradientDescent(learning_rate=0.05)
  tasks, samples = 8, 16
  cf = torch.randn(tasks, samples, 6, device="cuda")
  cl = torch.randn(tasks, samples, 2, device="cuda")
  inf = torch.randn(tasks, samples, 6, device="cuda")
  infl = torch.randn(tasks, samples, 2, device="cuda")

  # Loop path (2 adaptation steps per task).
  per_task_cond = []
  losses0 = []
  for t in range(tasks):
    f = tsu.TensorSpecStruct(); f["x"] = cf[t]
    l = tsu.TensorSpecStruct(); l["y"] = cl[t]
    vf = tsu.TensorSpecStruct(); vf["x"] = inf[t]
    vl = tsu.TensorSpecStruct(); vl["y"] = infl[t]
    (uncond, cond), _, inner_losses = loop.inner_loop(
        [(f, l), (f, l), (vf, vl)], base.inference_network_fn,
        base.model_train_fn, net, mode="train")
    per_task_cond.append(cond["prediction"])
    losses0.append(inner_losses[0])
  loop_cond = torch.stack(per_task_cond)
  loop_outer = loop_cond.pow(2).mean()
  loop_outer.backward()
  loop_grads = [p.grad.clone() for p in net.parameters()]
  for p in net.parameters():
    p.grad = None

  # vmap path.
  uncond, cond, inner_outs, losses = loop.inner_loop_vmapped(
      {"x": cf}, {"y": cl}, {"x": inf}, {"y": infl},
      {"cond_f": {}, "cond_l": {}, "inf_f": {}, "inf_l": {}},
      base.inference_network_fn, base.model_train_fn, net,
      num_steps=2, mode="train")
  torch.testing.assert_close(cond["prediction"], loop_cond,
                             rtol=1e-4, atol=1e-5)
  torch.testing.assert_close(losses[:, 0], torch.stack(losses0),
                             rtol=1e-4, atol=1e-6)
  vmap_outer = cond["prediction"].pow(2).mean()
  vmap_outer.backward()
  for g_loop, p in zip(loop_grads, net.parameters()):
    torch.testing.assert_close(p.grad, g_loop, rtol=1e-3, atol=1e-5)


@requires_gpu
def test_jpeg_restart_stream_gpu_batch_decode():
  """Restart-marker streams (segment-parallel Huffman) through the GPU
  decode path match the plain-stream decode bit-for-bit."""
  from tensor2robot_amd.data import gpu_jpeg
  from tensor2robot_amd.data import image_codec

  rng = np.random.default_rng(3)
  imgs = [(rng.random((96, 128, 3)) * 255).astype(np.uint8)
          for _ in range(6)]
  plain = [image_codec.encode_jpeg(im, quality=92) for im in imgs]
  rst = [image_codec.encode_jpeg(im, quality=92, restart_interval=16)
         for im in imgs]
  out_plain = gpu_jpeg.decode_jpeg_batch(plain, device="cuda")
  out_rst = gpu_jpeg.decode_jpeg_batch(rst, device="cuda")
  assert out_plain.shape == (6, 96, 128, 3)
  torch.testing.assert_close(out_rst, out_plain, rtol=0, atol=0)
