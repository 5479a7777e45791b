"""BC-Z: configurable behavior-cloning regression model.

Reference `research/bcz/model.py`: BCZPreprocessor :69 (jpeg parse ->
crop -> resize -> distortions, mixup :84, gripper binarize/rescale),
spatial_softmax_network :198, resnet_film_network :245,
predict_stop_network :289, infer_outputs :319 (residual xyz, quaternion
normalize + multiply, sigmoid gripper), training_outputs :476 (per-
component weighted huber/mse/log losses, quaternion-norm penalty :631,
stop-state CE :462-474), gripper metrics :588, BCZModel :641
(ConditionMode one-hot task id / 512-d language embedding :63,
task-embedding noise :814, state/past conditioning :823-860).

MI355X design: the network is a torch module (ResNet-FiLM runs
channels_last with the fused CDNA4 BN+ReLU kernels from ops/fused_bn);
preprocessing (crop/resize/distort) runs on-GPU after H2D of raw uint8.
"""

from __future__ import annotations

import enum
from typing import Dict, List, Optional, Sequence

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin
from tensor2robot_amd.layers import bcz_networks
from tensor2robot_amd.layers import resnet as resnet_mod
from tensor2robot_amd.layers import vision_layers
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.preprocessors import abstract_preprocessor
from tensor2robot_amd.preprocessors import distortion
from tensor2robot_amd.research.bcz import pose_components as pose_lib
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes

TSPEC = tsu.ExtendedTensorSpec
TRAIN = run_modes.TRAIN

NUM_DEBUG_TASKS = 21  # reference :55
GRIPPER_CLOSE_FRACTION_TO_OPEN_GRIPPER = 0.4  # reference :58
MIN_GRIPPER_CLOSE = 0.2  # reference :59


@gin.constants_from_enum
class ConditionMode(enum.Enum):
  ONEHOT_TASKID = 1
  LANGUAGE_EMBEDDING = 2


# ---------------------------------------------------------------- losses
def weighted_loss(elementwise: torch.Tensor,
                  weights: torch.Tensor) -> torch.Tensor:
  """TF tf.losses SUM_BY_NONZERO_WEIGHTS reduction."""
  weights = torch.as_tensor(weights, dtype=elementwise.dtype,
                            device=elementwise.device)
  weights = weights.expand_as(elementwise)
  num = (weights != 0).sum().clamp(min=1).to(elementwise.dtype)
  return (elementwise * weights).sum() / num


def huber(labels, predictions, weights, delta: float = 1.0):
  return weighted_loss(
      F.huber_loss(predictions, labels, reduction="none", delta=delta),
      weights)


def mse(labels, predictions, weights):
  return weighted_loss((predictions - labels) ** 2, weights)


def log_loss(labels, predictions, weights, eps: float = 1e-7):
  p = predictions.clamp(eps, 1.0 - eps)
  return weighted_loss(-labels * torch.log(p)
                       - (1.0 - labels) * torch.log(1.0 - p), weights)


@gin.configurable
def piecewise_scaled_huber(labels, predictions, weights,
                           threshold: float = 0.2, slope: float = 0.001):
  """Scale down very large losses instead of clipping (reference :631-638)."""
  loss = huber(labels, predictions, weights)
  return torch.where(loss > 1.0, threshold + (loss - threshold) * slope,
                     loss)


def clipped_huber(labels, predictions, weights):
  return huber(labels, predictions, weights).clamp(0.0, 6.0)  # relu6


_LOSS_FNS = {
    "mse": mse,
    "huber": huber,
    "clipped_huber": clipped_huber,
    "piecewise_scaled_huber": piecewise_scaled_huber,
}


@gin.configurable
def compute_stop_state_loss(stop_state_labels: torch.Tensor,
                            stop_state_predictions: torch.Tensor,
                            class_weights: Sequence[float] = (1.0, 1.0, 1.0)
                            ) -> torch.Tensor:
  """Weighted softmax CE over (continue, fail/help, success) (ref :462-474)."""
  cw = torch.as_tensor(class_weights,
                       dtype=stop_state_predictions.dtype,
                       device=stop_state_predictions.device)
  weights = (stop_state_labels * cw).sum(-1)
  ce = -(stop_state_labels
         * torch.log_softmax(stop_state_predictions, dim=-1)).sum(-1)
  return weighted_loss(ce, weights)


def quaternion_multiply(q1: torch.Tensor, q2: torch.Tensor) -> torch.Tensor:
  """Hamilton product, (x, y, z, w) layout (reference quaternion_lib)."""
  x1, y1, z1, w1 = q1.unbind(-1)
  x2, y2, z2, w2 = q2.unbind(-1)
  return torch.stack([
      w1 * x2 + x1 * w2 + y1 * z2 - z1 * y2,
      w1 * y2 - x1 * z2 + y1 * w2 + z1 * x2,
      w1 * z2 + x1 * y2 - y1 * x2 + z1 * w2,
      w1 * w2 - x1 * x2 - y1 * y2 - z1 * z2,
  ], dim=-1)


# ---------------------------------------------------------- preprocessor
@gin.configurable
class BCZPreprocessor(abstract_preprocessor.SpecTransformationPreprocessor):
  """Image conversion, crop, resize, distortion, mixup (reference :69-195)."""

  def __init__(self, image_size=(100, 100), crop_size=(512, 640),
               input_size=(512, 640), is_sequence: bool = False,
               mixup_alpha: float = 0.0, binarize_gripper: bool = True,
               rescale_gripper: bool = False, mock_subtask: bool = False,
               **kwargs):
    self._image_size = tuple(image_size)
    self._crop_size = tuple(crop_size)
    self._input_size = tuple(input_size)
    self._is_sequence = is_sequence
    self._mixup_alpha = mixup_alpha
    self._binarize_gripper = binarize_gripper
    self._rescale_gripper = rescale_gripper
    self._mock_subtask = mock_subtask
    super().__init__(**kwargs)

  @property
  def rescale_gripper(self) -> bool:
    return self._rescale_gripper

  def get_in_feature_specification(self, mode):
    flat = tsu.flatten_spec_structure(
        self.model_feature_specification_fn(mode))
    out = tsu.TensorSpecStruct()
    for key, spec in flat.items():
      # original_image is produced BY preprocessing (reference :104-113).
      if mode != run_modes.PREDICT and key in ("original_image",
                                               "original_depth_image"):
        continue
      if key == "image":
        out[key] = TSPEC(self._input_size + (3,), torch.uint8,
                         name=spec.name, data_format="JPEG",
                         dataset_key=spec.dataset_key)
      else:
        out[key] = spec
    return out

  def _preprocess_fn(self, features, labels, mode):
    features["original_image"] = features["image"]
    features["image"] = distortion.preprocess_image(
        features["image"], mode, is_sequence=self._is_sequence,
        input_size=self._input_size, target_size=self._image_size,
        crop_size=self._crop_size)
    if self._mixup_alpha > 0.0 and labels is not None and mode == TRAIN:
      # Mixup against the batch-reversed pairing (reference :166-173).
      m = torch.distributions.Beta(self._mixup_alpha,
                                   self._mixup_alpha).sample().item()
      img = features["image"]
      features["image"] = m * img + (1 - m) * torch.flip(img, dims=[0])
      for key in list(labels.keys()):
        if key.startswith("future/"):
          x = labels[key]
          labels[key] = m * x + (1 - m) * torch.flip(x, dims=[0])
    key = "future/target_close"
    if labels is not None and self._binarize_gripper and key in labels:
      labels[key] = (labels[key]
                     > GRIPPER_CLOSE_FRACTION_TO_OPEN_GRIPPER).to(
                         labels[key].dtype)
    if labels is not None and self._rescale_gripper and key in labels:
      labels[key] = ((labels[key] - MIN_GRIPPER_CLOSE)
                     / (1 - MIN_GRIPPER_CLOSE)).clamp(min=0.0)
    if self._mock_subtask and "subtask_id" in features:
      features["subtask_id"] = torch.zeros_like(features["subtask_id"])
    return features, labels


# -------------------------------------------------------------- networks
@gin.configurable
class SpatialSoftmaxNetwork(nn.Module):
  """Berkeley-Net torso + pose MLP (reference spatial_softmax_network :198)."""

  def __init__(self, pose_components, num_waypoints: int,
               condition_dim: int = 0, in_channels: int = 3):
    super().__init__()
    self.pose_components = list(pose_components)
    self.num_waypoints = num_waypoints
    self.torso = vision_layers.ImagesToFeaturesNet(
        in_channels=in_channels, normalizer="layer")
    action_sizes = [c[1] for c in self.pose_components]
    self.pose_net = vision_layers.ImageFeaturesToPoseNet(
        feature_dim=64 + condition_dim,
        num_outputs=sum(action_sizes) * num_waypoints,
        bias_transform_size=10)

  def forward(self, image: torch.Tensor,
              condition_input: Optional[torch.Tensor] = None):
    feature_points, _ = self.torso(image)
    if condition_input is not None:
      feature_points = torch.cat([feature_points, condition_input], dim=-1)
    estimated_pose, _ = self.pose_net(feature_points)
    outputs: Dict[str, torch.Tensor] = {}
    i = 0
    for name, size, is_residual, _ in self.pose_components:
      if is_residual:
        name += "_residual"
      n = size * self.num_waypoints
      outputs[name] = estimated_pose[..., i:i + n].reshape(
          -1, self.num_waypoints, size)
      i += n
    return outputs, feature_points


@gin.configurable
class ResNetFiLMNetwork(nn.Module):
  """ResNet + FiLM conditioning + MultiHeadMLP (reference :245-287).

  block_layer3's pooled features are the state embedding used by the
  stop-state head (reference :283-284).
  """

  def __init__(self, pose_components, num_waypoints: int,
               condition_dim: int = 0, resnet_size: int = 18,
               fc_layers: Sequence[int] = (100, 100),
               enabled_block_layers: Optional[Sequence[bool]] = None,
               in_channels: int = 3):
    super().__init__()
    self.pose_components = list(pose_components)
    self.num_waypoints = num_waypoints
    self.resnet = resnet_mod.ResNet(resnet_size=resnet_size, num_classes=0,
                                    in_channels=in_channels)
    self.film_gen = None
    if condition_dim > 0:
      self.film_gen = resnet_mod.LinearFiLMGenerator(
          condition_dim, self.resnet,
          enabled_block_layers=enabled_block_layers)
    action_sizes = [c[1] for c in self.pose_components]
    self.heads = bcz_networks.MultiHeadMLP(
        self.resnet.out_channels, action_sizes, num_waypoints,
        fc_layers)
    expansion = 4 if resnet_size >= 50 else 1
    self.state_dim = 256 * expansion

  def forward(self, image: torch.Tensor,
              condition_input: Optional[torch.Tensor] = None):
    gbs = None
    if self.film_gen is not None and condition_input is not None:
      gbs = self.film_gen(condition_input)
    net, endpoints = self.resnet(image, film_gamma_betas=gbs)
    components = self.heads(net)
    outputs: Dict[str, torch.Tensor] = {}
    for (name, _, is_residual, _), comp in zip(self.pose_components,
                                               components):
      if is_residual:
        name += "_residual"
      outputs[name] = comp
    outputs["policy_image_features"] = net
    state_features = endpoints["block_layer3"].mean(dim=(2, 3))
    return outputs, state_features


class PredictStopNetwork(nn.Module):
  """3-way (continue, fail/help, success) head (reference :289-317)."""

  def __init__(self, in_dim: int, fc_layers: Sequence[int] = (100, 100),
               num_waypoints: int = 1):
    super().__init__()
    self.num_waypoints = num_waypoints
    layers: List[nn.Module] = []
    d = in_dim
    for width in fc_layers:
      layers += [nn.Linear(d, width), nn.LayerNorm(width), nn.ReLU()]
      d = width
    self.stack = nn.Sequential(*layers)
    self.head = nn.Linear(d, 3)
    self.rest_head = nn.Linear(d, (num_waypoints - 1) * 3) \
        if num_waypoints > 1 else None

  def forward(self, state_embedding: torch.Tensor) -> torch.Tensor:
    net = self.stack(state_embedding)
    logits = self.head(net)
    if self.rest_head is not None:
      rest = self.rest_head(net.detach() if self.training else net)
      logits = torch.cat([logits, rest], dim=-1)
    return logits


# ---------------------------------------------------- infer / train fns
def infer_outputs(features, network_output_dict, action_components,
                  rescale_target_close: bool) -> Dict[str, torch.Tensor]:
  """Network outputs -> absolute-pose inference outputs (reference :319-464)."""
  inference_outputs: Dict[str, torch.Tensor] = {}
  action_outputs = []
  for name, _, is_residual, _ in action_components:
    predict_name = name + "_residual" if is_residual else name
    value = network_output_dict[predict_name]
    if name == "xyz":
      if is_residual:
        value = value + features["present/xyz"].reshape(
            -1, 1, value.shape[-1])
      action_outputs.append(value)
    elif name == "quaternion":
      norm = torch.linalg.norm(value, dim=-1, keepdim=True)
      value = value / norm
      if is_residual:
        curr = features["present/quaternion"].reshape(-1, 1, 4)
        value = quaternion_multiply(curr, value)
      action_outputs.append(value)
      # Losses regress the normalized quaternion (reference :396-401).
      network_output_dict["quaternion"] = value
      inference_outputs["quaternion_norm"] = norm
    elif name in ("target_close", "stop_token"):
      if is_residual:
        raise ValueError(f"{name} does not support residual")
      value = torch.sigmoid(value)
      if rescale_target_close:
        value = MIN_GRIPPER_CLOSE + value * (1 - MIN_GRIPPER_CLOSE)
      action_outputs.append(value)
    elif name == "base_joystick_xy":
      action_outputs.append(torch.tanh(value))
    else:
      # xyz-like additive components (axis_angle, arm_joints, pantilt,
      # velocities — reference :403-434).
      if is_residual:
        curr = features["present/" + name].reshape(
            -1, 1, value.shape[-1])
        value = value + curr
      action_outputs.append(value)
  inference_outputs.update(network_output_dict)
  for (name, _, _, _), out in zip(action_components, action_outputs):
    inference_outputs["action/" + name] = out
  inference_outputs["action_trajectory"] = torch.cat(action_outputs,
                                                     dim=-1)
  if "image" in features:
    inference_outputs["image"] = features["image"]
  return inference_outputs


@gin.configurable
def training_outputs(features, labels, network_output_dict,
                     action_components, quaternion_penalty: float = 0.01,
                     loss_name: str = "huber"):
  """Per-component weighted losses (reference :476-586)."""
  del features
  reg_loss_fn = _LOSS_FNS[loss_name]
  stop_key = "future/stop_token"
  if stop_key in labels:
    stop_mask_value = 1.0 - labels[stop_key]
  else:
    stop_mask_value = None
  train_outputs: Dict[str, torch.Tensor] = {}
  nonloss: Dict[str, torch.Tensor] = {}
  for name, _, is_residual, weight in action_components:
    key = name + "_residual" if is_residual else name
    predicted = network_output_dict[key]
    label = labels["future/" + key]
    if name in ("target_close", "stop_token"):
      predicted = torch.sigmoid(predicted)
      nonloss[name + "_predicted"] = predicted
      loss_fn = log_loss
    else:
      loss_fn = reg_loss_fn
    if stop_mask_value is not None:
      mask = stop_mask_value * torch.ones_like(predicted)
    else:
      mask = torch.ones_like(predicted)
    train_outputs[name + "_loss"] = loss_fn(label, predicted,
                                            weight * mask)
    nonloss["first_" + name + "_error"] = loss_fn(
        label[..., 0, :], predicted[..., 0, :],
        torch.full_like(predicted[..., 0, :], weight))
  if "quaternion_norm" in network_output_dict:
    predicted = network_output_dict["quaternion_norm"]
    w = quaternion_penalty * (stop_mask_value
                              if stop_mask_value is not None else 1.0)
    train_outputs["quaternion_norm_loss"] = reg_loss_fn(
        torch.ones_like(predicted), predicted,
        w * torch.ones_like(predicted))
  if "stop_state" in network_output_dict:
    stop_labels = F.one_hot(labels["future/stop_state"].long(),
                            num_classes=3).float()
    train_outputs["stop_state_loss"] = compute_stop_state_loss(
        stop_labels, network_output_dict["stop_state"])
  loss = sum(train_outputs.values())
  train_outputs.update(nonloss)
  return loss, train_outputs


def get_gripper_accuracy_metrics(inference_outputs, features, labels):
  """Gripper open/close accuracy metrics (reference :588-620)."""
  key = "target_close"
  current = features["present/" + key]
  future = labels["future/" + key]
  pred = inference_outputs[key][:, 0]
  label0 = future[:, 0]
  metrics = {}
  for s, lbl, prd in [
      ("closing", (label0 - current > 0).float(),
       (pred - current > 0).float()),
      ("opening", (label0 - current < 0).float(),
       (pred - current < 0).float())]:
    tp = (prd * lbl).sum()
    metrics[s + "_accuracy"] = (prd == lbl).float().mean()
    metrics[s + "_precision"] = tp / prd.sum().clamp(min=1.0)
    metrics[s + "_recall"] = tp / lbl.sum().clamp(min=1.0)
    metrics[s + "_pos_freq"] = lbl.mean()
  return metrics


# ----------------------------------------------------------------- model
@gin.configurable
class BCZModel(abstract_model.AbstractT2RModel):
  """Single-image configurable BC regression model (reference :641-950)."""

  def __init__(self,
               state_components=None,
               action_components=None,
               predict_stop: bool = False,
               image_size=(100, 100),
               input_size=None,
               num_waypoints: int = 1,
               num_past: int = 0,
               num_total_users: int = 0,
               network: str = "resnet_film",
               resnet_size: int = 18,
               ignore_task_embedding: bool = False,
               task_embedding_noise_std: float = 0.1,
               mask_stop_token: bool = False,
               cond_modality: ConditionMode = ConditionMode.ONEHOT_TASKID,
               **kwargs):
    kwargs.setdefault("preprocessor_cls", BCZPreprocessor)
    super().__init__(**kwargs)
    self._image_size = tuple(image_size)
    self._input_size = tuple(input_size) if input_size else (512, 640)
    self._predict_stop = predict_stop
    self._num_waypoints = num_waypoints
    self._num_past = num_past
    self._num_total_users = num_total_users
    self._network_kind = network
    self._resnet_size = resnet_size
    self._ignore_task_embedding = ignore_task_embedding
    self._task_embedding_noise_std = task_embedding_noise_std
    self._mask_stop_token = mask_stop_token
    self._cond_mode = cond_modality
    self._action_components = list(
        action_components
        if action_components is not None
        else pose_lib.DEFAULT_ACTION_COMPONENTS)
    self._state_components = list(state_components or [])

  @property
  def default_preprocessor_cls(self):
    return BCZPreprocessor

  @property
  def action_component_names(self):
    return [c[0] for c in self._action_components]

  @property
  def is_joint_space(self):
    return "arm_joints" in self.action_component_names

  @property
  def is_xyz_space(self):
    return "xyz" in self.action_component_names

  def get_feature_specification(self, mode):
    del mode
    features = tsu.TensorSpecStruct()
    features["image"] = TSPEC(
        self._image_size + (3,), torch.float32,
        name="present/image/encoded", data_format="jpeg")
    for name, size, _ in self._state_components:
      features["present/" + name] = TSPEC((size,), torch.float32,
                                          name="present/" + name)
    for name, size, _, _ in self._action_components:
      data_name = "sensed_close" if name == "target_close" else name
      features["present/" + name] = TSPEC((size,), torch.float32,
                                          name="present/" + data_name)
    if self._cond_mode == ConditionMode.ONEHOT_TASKID:
      features["subtask_id"] = TSPEC((1,), torch.int64, name="subtask_id")
    else:
      features["sentence_embedding"] = TSPEC((512,), torch.float32,
                                             name="sentence_embedding")
    if self._num_total_users:
      features["user_id"] = TSPEC((1,), torch.int64, name="user_int")
    features["original_image"] = TSPEC(
        self._input_size + (3,), torch.uint8, data_format="jpeg",
        name="original_image", is_optional=True)
    if self._num_past:
      for name, size, residual in self._state_components:
        if residual:
          name += "_residual"
        features["past/" + name] = TSPEC((self._num_past, size),
                                         torch.float32,
                                         name="past/" + name)
    return features

  def get_label_specification(self, mode):
    del mode
    labels = tsu.TensorSpecStruct()
    if self._predict_stop:
      labels["future/stop_state"] = TSPEC((), torch.int64,
                                          name="present/stop_state")
    for name, size, residual, _ in self._action_components:
      if residual:
        name += "_residual"
      labels["future/" + name] = TSPEC((self._num_waypoints, size),
                                       torch.float32,
                                       name="future/" + name)
    if self._mask_stop_token:
      labels["future/stop_token"] = TSPEC((self._num_waypoints, 1),
                                          torch.float32,
                                          name="future/stop_token")
    return labels

  # -- conditioning --------------------------------------------------------
  def _condition_dim(self) -> int:
    if self._ignore_task_embedding:
      dim = 0
    elif self._cond_mode == ConditionMode.ONEHOT_TASKID:
      dim = NUM_DEBUG_TASKS
    else:
      dim = 512
    dim += sum(c[1] for c in self._state_components)
    if self._num_total_users:
      dim += self._num_total_users
    if self._num_past:
      dim += self._num_past * sum(c[1] for c in self._state_components)
    return dim

  def augment_condition_input(self, condition_input, features,
                              is_training: bool):
    """Noise + state/user/past concat (reference :823-860)."""
    if condition_input is not None and is_training and \
        self._task_embedding_noise_std:
      condition_input = condition_input + torch.randn_like(
          condition_input) * self._task_embedding_noise_std
    if self._ignore_task_embedding:
      condition_input = None
    parts = [] if condition_input is None else [condition_input]
    if self._state_components:
      parts.append(torch.cat(
          [features["present/" + c[0]] for c in self._state_components],
          dim=-1))
    if self._num_total_users:
      parts.append(F.one_hot(features["user_id"][:, 0].long(),
                             self._num_total_users).float())
    if self._num_past:
      prev = []
      for name, _, residual in self._state_components:
        if residual:
          name += "_residual"
        prev.append(features["past/" + name])
      prev = torch.cat(prev, dim=-1)
      parts.append(prev.reshape(prev.shape[0], -1))
    if not parts:
      return None
    return torch.cat(parts, dim=-1) if len(parts) > 1 else parts[0]

  # -- network -------------------------------------------------------------
  def create_network(self):
    cond_dim = self._condition_dim()
    if self._network_kind == "resnet_film":
      net = ResNetFiLMNetwork(self._action_components,
                              self._num_waypoints,
                              condition_dim=cond_dim,
                              resnet_size=self._resnet_size)
      state_dim = net.state_dim
    elif self._network_kind == "spatial_softmax":
      net = SpatialSoftmaxNetwork(self._action_components,
                                  self._num_waypoints,
                                  condition_dim=cond_dim)
      state_dim = 64 + cond_dim
    else:
      raise ValueError(f"unknown network {self._network_kind!r}")
    modules = {"policy": net}
    if self._predict_stop:
      modules["stop"] = PredictStopNetwork(
          state_dim, num_waypoints=self._num_waypoints)
    return nn.ModuleDict(modules)

  def inference_network_fn(self, features, labels, mode, params=None):
    is_training = mode == TRAIN
    if self._cond_mode == ConditionMode.ONEHOT_TASKID:
      condition_input = F.one_hot(features["subtask_id"][:, 0].long(),
                                  NUM_DEBUG_TASKS).float()
    else:
      condition_input = features["sentence_embedding"]
    condition_input = self.augment_condition_input(condition_input,
                                                   features, is_training)
    image = features["image"]
    if image.dim() == 4 and image.shape[-1] == 3:
      image = image.permute(0, 3, 1, 2).contiguous()
    network_outputs, state_embedding = self.network["policy"](
        image, condition_input)
    outputs = infer_outputs(features, network_outputs,
                            self._action_components,
                            self.preprocessor.rescale_gripper)
    if self._predict_stop:
      outputs["stop_state"] = self.network["stop"](state_embedding)
    if condition_input is not None:
      outputs["condition_input"] = condition_input
    return outputs

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    return training_outputs(features, labels, inference_outputs,
                            self._action_components)

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    """Streaming means of train outputs + stop/gripper metrics (ref :895)."""
    metrics = {}
    if train_outputs is not None:
      for key, value in train_outputs.items():
        if isinstance(value, torch.Tensor) and value.dim() >= 2:
          continue
        metrics["mean_" + key] = value
    if self._predict_stop:
      pred = inference_outputs["stop_state"][..., :3].argmax(-1)
      metrics["accuracy_stop_state"] = (
          pred == labels["future/stop_state"].long()).float().mean()
    if train_outputs is not None and labels is not None and \
        "target_close" in self.action_component_names:
      metrics.update(get_gripper_accuracy_metrics(
          inference_outputs, features, labels))
    return metrics


def xyz_action_trajectory(outputs):
  """Concats xyz + rotation action streams (reference model.py:621-627)."""
  if "action/quaternion" in outputs:
    rotation = outputs["action/quaternion"]
  elif "action/axis_angle" in outputs:
    rotation = outputs["action/axis_angle"]
  else:
    raise KeyError("outputs carry neither action/quaternion nor "
                   "action/axis_angle")
  return torch.cat([outputs["action/xyz"], rotation], dim=-1)
