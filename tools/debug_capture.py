"""Bisect which part of model_fn breaks hipGraph capture (segfault in
capture_end).  Run variants in SEPARATE processes (a segfault kills the
process): python tools/debug_capture.py <variant>
variants: net | netloss | infer | train_fn | packed | summaries | full
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tensor2robot_amd.research.qtopt import t2r_models
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes


def main(variant):
  from tensor2robot_amd.utils import miopen_db
  miopen_db.use_packaged_db()
  torch.backends.cudnn.benchmark = True
  device = torch.device("cuda:0")
  model = t2r_models.GraspingModel(device_type="gpu",
                                   compute_dtype="bfloat16",
                                   use_avg_model_params=True)
  model.to_device(device)
  network = model.network
  network.to(memory_format=torch.channels_last)
  optimizer = model.create_optimizer()
  ema = model.create_ema()

  bs = 8
  g = torch.Generator().manual_seed(0)
  features = tsu.TensorSpecStruct()
  features["state/image"] = torch.rand(
      (bs, 472, 472, 3), generator=g).to(device)
  act = torch.rand(bs, t2r_models.ACTION_DIM, generator=g).to(device)
  off = 0
  for name, size in t2r_models.ACTION_COMPONENTS:
    features["action/" + name] = act[:, off:off + size].contiguous()
    off += size
  labels = tsu.TensorSpecStruct()
  labels["grasp_success"] = (torch.rand(bs, 1, generator=g) > 0.5
                             ).float().to(device)

  autocast = torch.autocast("cuda", dtype=torch.bfloat16)

  def loss_net():
    image = features["state/image"].permute(0, 3, 1, 2).contiguous(
        memory_format=torch.channels_last)
    action = model.pack_action_vector(features)
    logit = network(image, action)
    return torch.nn.functional.binary_cross_entropy_with_logits(
        logit.float(), labels["grasp_success"].reshape(logit.shape))

  def loss_infer():
    out = model.inference_network_fn(features, labels, run_modes.TRAIN)
    return torch.nn.functional.binary_cross_entropy_with_logits(
        out["logit"].float(),
        labels["grasp_success"].reshape(out["logit"].shape))

  def loss_train_fn():
    out = model.inference_network_fn(features, labels, run_modes.TRAIN)
    return model.model_train_fn(features, labels, out, run_modes.TRAIN)

  def loss_packed():
    f = tsu.validate_and_pack(
        model.get_feature_specification_for_packing(run_modes.TRAIN),
        features, ignore_batch=True)
    l = tsu.validate_and_pack(
        model.get_label_specification_for_packing(run_modes.TRAIN),
        labels, ignore_batch=True)
    out = model.inference_network_fn(f, l, run_modes.TRAIN)
    return model.model_train_fn(f, l, out, run_modes.TRAIN)

  def loss_summaries():
    loss = loss_packed()
    model.add_summaries(features, labels, None, loss, None,
                        run_modes.TRAIN)
    model.pop_scalar_summaries()
    return loss

  def loss_full():
    return model.model_fn(features, labels, run_modes.TRAIN).loss

  loss_fn = {"net": loss_net, "netloss": loss_net, "infer": loss_infer,
             "train_fn": loss_train_fn, "packed": loss_packed,
             "summaries": loss_summaries, "full": loss_full}[variant]

  def body():
    optimizer.zero_grad(set_to_none=True)
    with autocast:
      loss = loss_fn()
    loss.backward()
    optimizer.step(0)
    ema.update()
    return loss

  for i in range(3):
    body()
  torch.cuda.synchronize()
  print(f"[{variant}] eager ok", flush=True)

  side = torch.cuda.Stream()
  side.wait_stream(torch.cuda.current_stream())
  with torch.cuda.stream(side):
    for _ in range(3):
      body()
  torch.cuda.current_stream().wait_stream(side)
  torch.cuda.synchronize()
  print(f"[{variant}] warmup ok", flush=True)

  graph = torch.cuda.CUDAGraph()
  with torch.cuda.graph(graph):
    out = body()
  print(f"[{variant}] capture ok", flush=True)
  graph.replay()
  torch.cuda.synchronize()
  print(f"[{variant}] replay ok loss={float(out):.4f}", flush=True)


def main_trainer(variant):
  """Reproduce the failing test path: Trainer + pool (+ preprocess)."""
  import functools
  import itertools
  from tensor2robot_amd.train import train_eval
  torch.manual_seed(0)
  model = t2r_models.GraspingModel(device_type="gpu",
                                   compute_dtype="bfloat16",
                                   use_avg_model_params=True)
  trainer = train_eval.Trainer(model, model_dir="")
  device = trainer.device
  bs = 8
  g = torch.Generator().manual_seed(0)
  pool = []
  for _ in range(2):
    f = tsu.TensorSpecStruct()
    f["state/image"] = torch.randint(
        0, 256, (bs, 512, 640, 3), generator=g,
        dtype=torch.uint8).to(device)
    off = 0
    act = torch.rand(bs, t2r_models.ACTION_DIM, generator=g)
    for name, size in t2r_models.ACTION_COMPONENTS:
      f["action/" + name] = act[:, off:off + size].to(device)
      off += size
    l = tsu.TensorSpecStruct()
    l["grasp_success"] = (torch.rand(bs, 1, generator=g) >
                          0.5).float().to(device)
    pool.append((f, l))
  preprocess_fn = functools.partial(model.preprocessor.preprocess,
                                    mode=run_modes.TRAIN)
  if variant == "trainer_nopre":
    with torch.no_grad():
      pool = [model.preprocessor.preprocess(f, l, run_modes.EVAL)
              for f, l in pool]
    preprocess_fn = None
  cyc = itertools.cycle(pool)
  trainer.train(lambda: cyc, max_steps=5, preprocess_fn=preprocess_fn)
  assert trainer._fast_engine is not None
  print(f"[{variant}] graphed={trainer._fast_engine.is_graphed}",
        flush=True)


if __name__ == "__main__":
  v = sys.argv[1]
  if v.startswith("trainer"):
    main_trainer(v)
  else:
    main(v)
