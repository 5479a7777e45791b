"""AbstractT2RModel: the user-facing model base class.

Reference: `models/abstract_model.py:162-936`.  A T2R model declares its
feature/label specs, an `inference_network_fn` producing inference outputs,
a `model_train_fn` producing the loss, and optionally `model_eval_fn`,
`add_summaries` and `create_export_outputs_fn`.  The framework composes
these through `model_fn` (reference :662-834) and drives everything else
(optimizer, EMA, checkpoints, hooks) from the native train loop
(`tensor2robot_amd/train/train_eval.py`).

MI355X-first differences from the TF1 original:
  * define-by-run: networks are torch.nn.Modules created once via
    `create_network()` and reused every step.
  * bf16 compute via autocast on the CUDA (ROCm/HIP) device; parameters stay
    f32 master copies (the TPU bf16-rewrite machinery
    `tpu_model_wrapper.py:107-125` collapses into `compute_dtype`).
  * distribution is handled outside the model: the train loop wraps the
    network in the framework's bucketed-RCCL data-parallel engine
    (`tensor2robot_amd/parallel/ddp.py`) — the CrossShardOptimizer analog.
"""

from __future__ import annotations

import abc
import collections
from typing import Any, Callable, Dict, Optional

import torch

from tensor2robot_amd import gin
from tensor2robot_amd.models import model_interface
from tensor2robot_amd.models import optimizers as optimizers_mod
from tensor2robot_amd.preprocessors import abstract_preprocessor
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes

TRAIN = run_modes.TRAIN
EVAL = run_modes.EVAL
PREDICT = run_modes.PREDICT

DEVICE_TYPE_CPU = "cpu"
DEVICE_TYPE_GPU = "gpu"


class ModelFnOps(
    collections.namedtuple(
        "ModelFnOps",
        ["loss", "inference_outputs", "predictions", "train_outputs",
         "metrics", "scalar_summaries"])):
  """What one composed model_fn invocation produced."""


@gin.configurable
class AbstractT2RModel(model_interface.ModelInterface):
  """Base class all models derive from (reference abstract_model.py:162)."""

  def __init__(self,
               preprocessor_cls=None,
               create_optimizer_fn=optimizers_mod.default_create_optimizer_fn,
               device_type: str = DEVICE_TYPE_GPU,
               summarize_gradients: bool = False,
               use_avg_model_params: bool = False,
               avg_model_params_decay: float = 0.9999,
               init_from_checkpoint_fn: Optional[Callable] = None,
               use_sync_replicas_optimizer: bool = False,
               compute_dtype: str = "bfloat16"):
    # use_sync_replicas_optimizer (reference abstract_model.py:201-203,
    # 864-870): the reference's PS-architecture SyncReplicasOptimizer
    # collapses into the always-synchronous RCCL all-reduce of the DP
    # engine here — the flag is accepted for config parity and noted,
    # but every multi-process run IS synchronous data-parallel.
    self._use_sync_replicas_optimizer = use_sync_replicas_optimizer
    self._preprocessor_cls = preprocessor_cls
    self._create_optimizer_fn = create_optimizer_fn
    if device_type not in (DEVICE_TYPE_CPU, DEVICE_TYPE_GPU):
      raise ValueError(f"Unknown device_type {device_type}")
    self._device_type = device_type
    self._summarize_gradients = summarize_gradients
    self._use_avg_model_params = use_avg_model_params
    self._avg_model_params_decay = avg_model_params_decay
    self._init_from_checkpoint_fn = init_from_checkpoint_fn
    self._compute_dtype = tsu.canonical_dtype(compute_dtype)
    self._preprocessor = None
    self._network: Optional[torch.nn.Module] = None
    self._device = torch.device("cpu")
    self._scalar_summaries: Dict[str, float] = {}

  # -- specs (abstract) ----------------------------------------------------
  @abc.abstractmethod
  def get_feature_specification(self, mode) -> tsu.TensorSpecStruct:
    pass

  @abc.abstractmethod
  def get_label_specification(self, mode) -> tsu.TensorSpecStruct:
    pass

  # -- preprocessor --------------------------------------------------------
  @property
  def default_preprocessor_cls(self):
    return abstract_preprocessor.NoOpPreprocessor

  @property
  def preprocessor(self):
    if self._preprocessor is None:
      preprocessor_cls = self._preprocessor_cls or \
          self.default_preprocessor_cls
      self._preprocessor = preprocessor_cls(
          model_feature_specification_fn=self.get_feature_specification,
          model_label_specification_fn=self.get_label_specification)
    return self._preprocessor

  @property
  def device_type(self) -> str:
    return self._device_type

  @property
  def device(self) -> torch.device:
    return self._device

  @property
  def compute_dtype(self) -> torch.dtype:
    return self._compute_dtype

  @property
  def use_avg_model_params(self) -> bool:
    return self._use_avg_model_params

  @property
  def avg_model_params_decay(self) -> float:
    return self._avg_model_params_decay

  @property
  def init_from_checkpoint_fn(self):
    return self._init_from_checkpoint_fn

  # -- network lifecycle ---------------------------------------------------
  @abc.abstractmethod
  def create_network(self) -> torch.nn.Module:
    """Creates the torch module(s); called once, lazily."""

  @property
  def network(self) -> torch.nn.Module:
    if self._network is None:
      self._network = self.create_network()
      self._network.to(self._device)
      if self._init_from_checkpoint_fn is not None:
        self._init_from_checkpoint_fn(self._network)
    return self._network

  def to_device(self, device) -> "AbstractT2RModel":
    self._device = torch.device(device)
    if self._network is not None:
      self._network.to(self._device)
    return self

  def set_network(self, network: torch.nn.Module):
    """Replaces the live network (used by the DP wrapper)."""
    self._network = network

  # -- the four model hooks ------------------------------------------------
  @abc.abstractmethod
  def inference_network_fn(self, features, labels, mode,
                           params=None) -> Dict[str, torch.Tensor]:
    """Forward pass -> inference outputs dict (reference :404-451)."""

  @abc.abstractmethod
  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    """Returns scalar loss or (loss, train_outputs) (reference :453-504)."""

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None) -> Dict[str, Any]:
    """Eval metric dict; default loss-only (reference :506-565)."""
    return {}

  def add_summaries(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    """Scalar summary hook-point (reference :567-608)."""
    del features, labels, inference_outputs, train_outputs, mode, params
    if train_loss is not None:
      self.scalar_summary("loss", train_loss)

  def create_export_outputs_fn(self, features, inference_outputs, mode,
                               params=None) -> Dict[str, torch.Tensor]:
    """Predictions exposed by exported/serving models (reference :714-736)."""
    del features, mode, params
    return dict(inference_outputs)

  def scalar_summary(self, name: str, value):
    # Keep tensors AS tensors: a .cpu() here is a device sync, which is
    # illegal inside hipGraph capture (the Trainer's graphed step runs
    # model_fn under capture) and a per-step stall otherwise.  The
    # Trainer materializes floats only at summary-write time.
    if isinstance(value, torch.Tensor):
      value = value.detach()
    self._scalar_summaries[name] = value

  def pop_scalar_summaries(self) -> Dict[str, float]:
    out = self._scalar_summaries
    self._scalar_summaries = {}
    return out

  # -- composition ---------------------------------------------------------
  def model_fn(self, features, labels, mode, params=None) -> ModelFnOps:
    """Composes the model hooks for one batch (reference :662-834)."""
    run_modes.validate(mode)
    features = tsu.validate_and_pack(
        self.get_feature_specification_for_packing(mode), features,
        ignore_batch=True)
    if labels is not None:
      labels = tsu.validate_and_pack(
          self.get_label_specification_for_packing(mode), labels,
          ignore_batch=True)
    inference_outputs = self.inference_network_fn(features, labels, mode,
                                                  params)
    if mode == PREDICT:
      predictions = self.create_export_outputs_fn(features,
                                                  inference_outputs, mode,
                                                  params)
      return ModelFnOps(loss=None, inference_outputs=inference_outputs,
                        predictions=predictions, train_outputs=None,
                        metrics=None, scalar_summaries={})

    train_fn_result = self.model_train_fn(features, labels,
                                          inference_outputs, mode, params)
    if isinstance(train_fn_result, torch.Tensor):
      train_loss, train_outputs = train_fn_result, None
    elif isinstance(train_fn_result, tuple) and len(train_fn_result) == 2:
      train_loss, train_outputs = train_fn_result
    else:
      raise ValueError(
          "model_train_fn must return loss or (loss, train_outputs)")

    metrics = None
    if mode == EVAL:
      metrics = self.model_eval_fn(features, labels, inference_outputs,
                                   train_loss, train_outputs, mode, params)
    self.add_summaries(features, labels, inference_outputs, train_loss,
                       train_outputs, mode, params)
    return ModelFnOps(loss=train_loss, inference_outputs=inference_outputs,
                      predictions=None, train_outputs=train_outputs,
                      metrics=metrics,
                      scalar_summaries=self.pop_scalar_summaries())

  # -- optimizer -----------------------------------------------------------
  def create_optimizer(self) -> optimizers_mod.ScheduledOptimizer:
    """Builds the optimizer over network params (reference :836-871)."""
    params = [p for p in self.network.parameters() if p.requires_grad]
    if not params:
      raise ValueError("Model has no trainable parameters")
    return self._create_optimizer_fn()(self.filter_trainables(params))

  def filter_trainables(self, params):
    """Overridable trainable-variable filter (reference :367-374)."""
    return params

  def create_ema(self) -> Optional[optimizers_mod.ExponentialMovingAverage]:
    if not self._use_avg_model_params:
      return None
    return optimizers_mod.ExponentialMovingAverage(
        self.network, decay=self._avg_model_params_decay)


@gin.configurable
def default_init_from_checkpoint_fn(checkpoint: str = None,
                                    allow_partial_restore: bool = False,
                                    name_filter: Optional[str] = None):
  """Warm-start fn factory: partial restore with filtering.

  Reference `abstract_model.py:86-126` (assignment-map warm start).
  """
  if checkpoint is None:
    raise ValueError("checkpoint must be provided")

  def init_fn(network: torch.nn.Module):
    payload = torch.load(checkpoint, map_location="cpu", weights_only=False)
    state = payload.get("model_state", payload)
    if name_filter:
      state = {k: v for k, v in state.items() if name_filter in k}
    missing, unexpected = network.load_state_dict(state, strict=False)
    if not allow_partial_restore and missing:
      raise ValueError(
          f"Restore from {checkpoint} missing keys: {missing}")
    return missing, unexpected

  return init_fn
