"""MAMLModel: wraps any base T2RModel into MAML meta-learning.

Reference `meta_learning/maml_model.py:71-...`: feature spec becomes
{condition: {features, labels}, inference: {features}} via
create_maml_feature_spec :126-137; inference_network_fn :229 maps the
per-task adaptation (`task_learn` :260) over the task dimension;
predictions expose full_condition_outputs/output_{i}, conditioned and
unconditioned inference outputs :318-343 with the subclass hook
_select_inference_output :356; model_train_fn :415 flattens
[tasks, samples] -> batch and computes the base model's outer loss on
post-adaptation inference outputs :466-496; create_train_op var_scope
filtering :373 maps to filter_trainables.

MI355X note: tasks iterate in Python (small task batches) but each
task's forwards are full-batch convs over [samples, ...] — the GPU sees
the same large launches as the base model; the inner update is pure
autograd (create_graph for second order).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
from torch import nn

from tensor2robot_amd import gin
from tensor2robot_amd.meta_learning import maml_inner_loop
from tensor2robot_amd.meta_learning import meta_tfdata
from tensor2robot_amd.meta_learning import preprocessors
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.specs import tensorspec_utils as tsu


def _index_struct(struct, i: int):
  out = tsu.TensorSpecStruct()
  for k, v in tsu.flatten_spec_structure(struct).items():
    out[k] = v[i] if isinstance(v, torch.Tensor) else v
  return out


def _stack_structs(structs: List):
  out = tsu.TensorSpecStruct()
  if not structs:
    return out
  for k in tsu.flatten_spec_structure(structs[0]).keys():
    vals = [tsu.flatten_spec_structure(s)[k] for s in structs]
    if isinstance(vals[0], torch.Tensor):
      out[k] = torch.stack(vals, dim=0)
    else:
      out[k] = vals[0]
  return out


@gin.configurable
class MAMLModel(abstract_model.AbstractT2RModel):
  """Model-agnostic meta-learning wrapper (reference :71)."""

  def __init__(self, base_model, preprocessor_cls=None,
               num_inner_loop_steps: int = 1,
               var_scope: Optional[str] = None,
               inner_learning_rate: float = 0.001,
               use_second_order: bool = True,
               learn_inner_lr: bool = False,
               inner_var_scope: Optional[str] = None,
               parallel_tasks: bool = False,
               **kwargs):
    super().__init__(**kwargs)
    self._base_model = base_model
    self._maml_preprocessor_cls = preprocessor_cls
    self._num_inner_loop_steps = max(1, num_inner_loop_steps)
    # Reference maml_model.py use_parallel_for: adapt all tasks at once
    # (vmap) instead of a Python loop.  Requires a base network whose
    # forward doesn't mutate buffers (e.g. no train-mode BatchNorm).
    self._parallel_tasks = parallel_tasks
    self._var_scope = var_scope
    self._inner_learning_rate = inner_learning_rate
    self._use_second_order = use_second_order
    self._learn_inner_lr = learn_inner_lr
    self._inner_var_scope = inner_var_scope
    self._inner_loop = maml_inner_loop.MAMLInnerLoopGradientDescent(
        learning_rate=inner_learning_rate,
        use_second_order=use_second_order,
        var_scope=inner_var_scope,
        learn_inner_lr=learn_inner_lr)

  @property
  def base_model(self):
    return self._base_model

  @property
  def default_preprocessor_cls(self):
    return preprocessors.MAMLPreprocessorV2

  @property
  def preprocessor(self):
    if self._preprocessor is None:
      cls = self._maml_preprocessor_cls or self.default_preprocessor_cls
      self._preprocessor = cls(
          base_preprocessor=self._base_model.preprocessor)
    return self._preprocessor

  def get_feature_specification(self, mode):
    return preprocessors.create_maml_feature_spec(
        self._base_model.get_feature_specification(mode),
        self._base_model.get_label_specification(mode))

  def get_label_specification(self, mode):
    return preprocessors.create_maml_label_spec(
        self._base_model.get_label_specification(mode))

  # -- network -------------------------------------------------------------
  def create_network(self):
    modules = {"base": self._base_model.network}
    if self._learn_inner_lr:
      modules["inner_lrs"] = self._inner_loop.create_inner_lr_params(
          self._base_model.network)
    return nn.ModuleDict(modules)

  def to_device(self, device):
    self._base_model.to_device(device)
    return super().to_device(device)

  # -- forward -------------------------------------------------------------
  def inference_network_fn(self, features, labels, mode, params=None):
    _ = self.network  # materialize base net + inner lrs
    cond_f = features["condition/features"]
    cond_l = features["condition/labels"]
    inf_f = features["inference/features"]
    unused_inference_labels = labels if labels is not None else cond_l

    num_tasks = next(iter(
        tsu.flatten_spec_structure(cond_f).values())).shape[0]
    if self._parallel_tasks:
      return self._inference_vmapped(cond_f, cond_l, inf_f,
                                     unused_inference_labels, num_tasks,
                                     mode, params)
    per_task_uncond, per_task_cond = [], []
    per_task_inner: List[List] = [
        [] for _ in range(self._num_inner_loop_steps + 1)]
    inner_loss_sums = [0.0] * (self._num_inner_loop_steps + 1)
    for t in range(num_tasks):
      cf, cl = _index_struct(cond_f, t), _index_struct(cond_l, t)
      inputs_list = [(cf, cl)] * self._num_inner_loop_steps + [
          (_index_struct(inf_f, t),
           _index_struct(unused_inference_labels, t))]
      (uncond, cond), inner_outputs, inner_losses = \
          self._inner_loop.inner_loop(
              inputs_list,
              inference_network_fn=self._base_model.inference_network_fn,
              model_train_fn=self._base_model.model_train_fn,
              network=self._base_model.network, mode=mode, params=params)
      per_task_uncond.append(uncond)
      per_task_cond.append(cond)
      for i, o in enumerate(inner_outputs):
        per_task_inner[i].append(o)
      for i, l in enumerate(inner_losses):
        inner_loss_sums[i] = inner_loss_sums[i] + l.detach()

    predictions = tsu.TensorSpecStruct()
    condition_stacked = [_stack_structs(s) for s in per_task_inner]
    for k, v in condition_stacked[0].items():
      predictions["full_condition_output/" + k] = v
    for pos, stacked in enumerate(condition_stacked):
      for k, v in stacked.items():
        predictions[f"full_condition_outputs/output_{pos}/" + k] = v
    for k, v in _stack_structs(per_task_uncond).items():
      predictions["full_inference_output_unconditioned/" + k] = v
    for k, v in _stack_structs(per_task_cond).items():
      predictions["full_inference_output/" + k] = v
    for i, s in enumerate(inner_loss_sums):
      # Keep the tensor: float() here is a device sync per step and
      # breaks hipGraph capture (scalar summaries materialize at
      # summary-write time in the Trainer).
      self.scalar_summary(
          f"inner_loss_{i}",
          s / num_tasks if isinstance(s, torch.Tensor)
          else float(s) / num_tasks)
    predictions = self._select_inference_output(predictions)
    if "condition_output" not in predictions:
      raise ValueError("The required condition_output is not in "
                       f"predictions {list(predictions.keys())}")
    if "inference_output" not in predictions:
      raise ValueError("The required inference_output is not in "
                       f"predictions {list(predictions.keys())}")
    return predictions

  def _inference_vmapped(self, cond_f, cond_l, inf_f, inf_l, num_tasks,
                         mode, params):
    """Task-parallel adaptation path (reference use_parallel_for).

    Splits each input struct into its tensor leaves (vmapped over the
    task dim) and non-tensor leaves (closed over), runs
    `inner_loop_vmapped`, and assembles the identical prediction keys
    the per-task Python loop produces — vmap's output stacking replaces
    `_stack_structs`.
    """
    def _split(struct):
      flat = tsu.flatten_spec_structure(struct)
      tensors = {k: v for k, v in flat.items()
                 if isinstance(v, torch.Tensor)}
      static = {k: v for k, v in flat.items()
                if not isinstance(v, torch.Tensor)}
      return tensors, static

    cf_t, cf_s = _split(cond_f)
    cl_t, cl_s = _split(cond_l)
    if_t, if_s = _split(inf_f)
    il_t, il_s = _split(inf_l)
    static = {"cond_f": cf_s, "cond_l": cl_s, "inf_f": if_s,
              "inf_l": il_s}
    uncond, cond, inner_outs, losses = self._inner_loop.inner_loop_vmapped(
        cf_t, cl_t, if_t, il_t, static,
        inference_network_fn=self._base_model.inference_network_fn,
        model_train_fn=self._base_model.model_train_fn,
        network=self._base_model.network,
        num_steps=self._num_inner_loop_steps, mode=mode, params=params)

    predictions = tsu.TensorSpecStruct()
    for k, v in inner_outs[0].items():
      predictions["full_condition_output/" + k] = v
    for pos, outs in enumerate(inner_outs):
      for k, v in outs.items():
        predictions[f"full_condition_outputs/output_{pos}/" + k] = v
    for k, v in uncond.items():
      predictions["full_inference_output_unconditioned/" + k] = v
    for k, v in cond.items():
      predictions["full_inference_output/" + k] = v
    loss_means = losses.detach().mean(dim=0)  # [num_steps + 1]
    for i in range(loss_means.shape[0]):
      self.scalar_summary(f"inner_loss_{i}", loss_means[i])
    predictions = self._select_inference_output(predictions)
    for required in ("condition_output", "inference_output"):
      if required not in predictions:
        raise ValueError(f"The required {required} is not in "
                         f"predictions {list(predictions.keys())}")
    return predictions

  def _select_inference_output(self, predictions):
    """Subclass hook (reference :356): pick condition/inference outputs."""
    return predictions

  # -- outer loss ----------------------------------------------------------
  def filter_trainables(self, params):
    if self._var_scope is None:
      return params
    named = [(n, p) for n, p in self.network.named_parameters()
             if n.startswith(self._var_scope)]
    keep = {id(p) for _, p in named}
    return [p for p in params if id(p) in keep]

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    inference_flat = meta_tfdata.flatten_batch_examples(
        inference_outputs["full_inference_output"])
    features_flat = meta_tfdata.flatten_batch_examples(
        features["inference/features"])
    labels_flat = meta_tfdata.flatten_batch_examples(labels) \
        if labels is not None else None
    params = dict(params or {})
    params["is_outer_loss"] = True
    return self._base_model.model_train_fn(
        features=features_flat, labels=labels_flat,
        inference_outputs=inference_flat, mode=mode, params=params)

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    inference_flat = meta_tfdata.flatten_batch_examples(
        inference_outputs["full_inference_output"])
    features_flat = meta_tfdata.flatten_batch_examples(
        features["inference/features"])
    labels_flat = meta_tfdata.flatten_batch_examples(labels) \
        if labels is not None else None
    return self._base_model.model_eval_fn(
        features_flat, labels_flat, inference_flat, train_loss,
        train_outputs, mode, params)


def pfor_map_fn(fn, elems):
  """map_fn over dim 0 via torch.func.vmap (reference maml_model.py:43-69
  wrapped tf pfor; same contract: slices of `elems` through `fn`,
  results stacked on dim 0).  fn must be vmap-compatible (no in-place
  buffer mutation)."""
  return torch.func.vmap(fn, randomness="different")(elems)
