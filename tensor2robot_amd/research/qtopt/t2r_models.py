"""QT-Opt grasping critic model + preprocessor (the benchmark workload).

Reference: `research/qtopt/t2r_models.py` — LegacyGraspingModelWrapper(
CriticModel) :62 (log loss :229-241, momentum optimizer + EMA via
optimizer_builder :25, default hparams batch 32 / momentum .9 / lr 1e-4 /
EMA .9999 :77-89), Grasping44E2EOpenCloseTerminateGripperStatusHeightToBottom
:312-401 (state = 472x472x3 image; action = world_vector(3) +
vertical_rotation(2) + open/close/terminate flags + gripper_closed +
height_to_bottom), DefaultGrasping44ImagePreprocessor :242 (512x640 uint8
-> crop 472x472 -> distort in train).
"""

from __future__ import annotations

from typing import Optional

import torch

from tensor2robot_amd import gin
from tensor2robot_amd.models import classification_model
from tensor2robot_amd.models import optimizers as optimizers_mod
from tensor2robot_amd.policies import policies as policies_mod
from tensor2robot_amd.preprocessors import abstract_preprocessor
from tensor2robot_amd.preprocessors import distortion
from tensor2robot_amd.research.qtopt import networks
from tensor2robot_amd.specs import tensorspec_utils as tsu

TSPEC = tsu.ExtendedTensorSpec

# Raw (serialized) and cropped (model-facing) geometry (reference :42-43).
RAW_HEIGHT, RAW_WIDTH = 512, 640
CROP_HEIGHT, CROP_WIDTH = 472, 472

ACTION_COMPONENTS = (
    ("world_vector", 3),
    ("vertical_rotation", 2),
    ("close_gripper", 1),
    ("open_gripper", 1),
    ("terminate_episode", 1),
    ("gripper_closed", 1),
    ("height_to_bottom", 1),
)
ACTION_DIM = sum(size for _, size in ACTION_COMPONENTS)


@gin.configurable
class DefaultGrasping44ImagePreprocessor(
    abstract_preprocessor.SpecTransformationPreprocessor):
  """512x640 uint8 jpeg -> 472x472 f32 crop (+ train distortions).

  Runs on the GPU right after H2D transfer: uint8 travels over PCIe/xGMI,
  the convert+crop+distort work rides HIP (reference :242-310 semantics).
  """

  def _transform_in_feature_specification(self, flat_spec, mode):
    out = tsu.TensorSpecStruct()
    for key, spec in flat_spec.items():
      if key.endswith("state/image"):
        out[key] = TSPEC((RAW_HEIGHT, RAW_WIDTH, 3), torch.uint8,
                         name=spec.name, data_format="JPEG",
                         dataset_key=spec.dataset_key)
      else:
        out[key] = spec
    return out

  def _preprocess_fn(self, features, labels, mode):
    image = features["state/image"]
    image = distortion.preprocess_image(
        image, mode, input_size=(RAW_HEIGHT, RAW_WIDTH),
        target_size=(CROP_HEIGHT, CROP_WIDTH))
    features["state/image"] = image
    return features, labels


def default_create_optimizer_fn():
  """QT-Opt default: momentum 0.9, lr 1e-4 (reference :77-89)."""
  return optimizers_mod.create_momentum_optimizer(learning_rate=1e-4,
                                                  momentum=0.9)


@gin.configurable
class GraspingModel(classification_model.CriticModel):
  """The QT-Opt grasping critic (log loss on grasp success)."""

  def __init__(self,
               create_optimizer_fn=default_create_optimizer_fn,
               use_avg_model_params: bool = True,
               avg_model_params_decay: float = 0.9999,
               action_batch_size: Optional[int] = None,
               **kwargs):
    kwargs.setdefault("preprocessor_cls",
                      DefaultGrasping44ImagePreprocessor)
    super().__init__(create_optimizer_fn=create_optimizer_fn,
                     use_avg_model_params=use_avg_model_params,
                     avg_model_params_decay=avg_model_params_decay,
                     action_batch_size=action_batch_size, **kwargs)

  def get_state_specification(self):
    spec = tsu.TensorSpecStruct()
    spec["image"] = TSPEC((CROP_HEIGHT, CROP_WIDTH, 3), torch.float32,
                          name="state/image", data_format="JPEG")
    return spec

  def get_action_specification(self):
    spec = tsu.TensorSpecStruct()
    for name, size in ACTION_COMPONENTS:
      spec[name] = TSPEC((size,), torch.float32, name=name)
    return spec

  def get_label_specification(self, mode):
    spec = tsu.TensorSpecStruct()
    spec["grasp_success"] = TSPEC((1,), torch.float32,
                                  name="grasp_success")
    return spec

  def create_network(self):
    return networks.Grasping44(action_dim=ACTION_DIM)

  def pack_action_vector(self, features) -> torch.Tensor:
    """Concatenates action components in spec order (reference :149)."""
    parts = [features["action/" + name] for name, _ in ACTION_COMPONENTS]
    action = torch.cat(parts, dim=-1)
    return action

  def q_func(self, features, mode, params=None):
    image = features["state/image"]
    if image.dim() == 4 and image.shape[-1] == 3:
      image = image.permute(0, 3, 1, 2)  # NHWC (wire) -> NCHW tower input
    image = image.contiguous(memory_format=torch.channels_last)
    action = self.pack_action_vector(features)
    logit = self.network(image, action)
    return {"q_predicted": torch.sigmoid(logit), "logit": logit}

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    """Sigmoid log loss on grasp success (reference :229-241)."""
    logit = inference_outputs["logit"]
    target = labels.grasp_success.reshape(logit.shape)
    return torch.nn.functional.binary_cross_entropy_with_logits(
        logit.float(), target.float())

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    q = inference_outputs["q_predicted"]
    target = labels.grasp_success.reshape(q.shape)
    pred = (q > 0.5).float()
    return {"accuracy": (pred == target).float().mean()}

  def create_export_outputs_fn(self, features, inference_outputs, mode,
                               params=None):
    return {"q_predicted": inference_outputs["q_predicted"]}


@gin.configurable
class Grasping44E2EOpenCloseTerminateGripperStatusHeightToBottom(
    GraspingModel):
  """Concrete benchmark model (name parity with reference :312)."""


@gin.configurable
class GraspingCEMPolicy(policies_mod.CEMPolicy):
  """CEM policy wired to the grasping critic's action components.

  The action vector splits into the `action/<name>` component feed the
  critic's feature spec declares (reference policies.py:133-166 pack_fn
  + networks.py:412-423 megabatch tiling).
  """

  def __init__(self, **kwargs):
    kwargs.setdefault("action_size", ACTION_DIM)
    super().__init__(**kwargs)

  def _split_action(self, action):
    import numpy as np
    feed = {}
    offset = 0
    for name, size in ACTION_COMPONENTS:
      feed["action/" + name] = np.asarray(
          action[..., offset: offset + size], np.float32)
      offset += size
    return feed


def pack_features_kuka_e2e(t2r_model, *policy_inputs):
  """Policy-input packing for the real Kuka E2E env — unimplemented in
  the reference too (t2r_models.py:46-57 raises NotImplementedError)."""
  del t2r_model, policy_inputs
  raise NotImplementedError
