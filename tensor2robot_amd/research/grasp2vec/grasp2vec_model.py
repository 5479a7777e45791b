"""Grasp2Vec: self-supervised object-centric scene embeddings.

Reference `research/grasp2vec/grasp2vec_model.py`: maybe_crop_images :45
(shared random crop window in train, center in eval),
Grasp2VecPreprocessor :76 (512x640 jpeg -> crop -> float -> random
flips), Grasp2VecModel :136 (pregrasp/postgrasp/goal -> scene & goal
ResNet-50 towers, embedding loss = NPairs/Triplet/arithmetic), and
`networks.py:24` Embedding (ResNet-50 spatial -> relu -> mean pool).

MI355X notes: the two scene images are batched through one tower pass
(reference :188-190) — one big conv launch instead of two half-sized
ones, which keeps the 256 CUs full; channels_last + fused BN kernels
come from layers/resnet.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin
from tensor2robot_amd.layers import resnet as resnet_mod
from tensor2robot_amd.models import abstract_model
from tensor2robot_amd.preprocessors import abstract_preprocessor
from tensor2robot_amd.research.grasp2vec import losses
from tensor2robot_amd.specs import tensorspec_utils as tsu
from tensor2robot_amd.utils import modes as run_modes

TSPEC = tsu.ExtendedTensorSpec
TRAIN = run_modes.TRAIN

RAW_SHAPE = (512, 640, 3)  # serialized image geometry (reference :105)


def maybe_crop_images(images, params, mode: str,
                      generator: Optional[torch.Generator] = None):
  """Shared crop window: random offsets in train, centered otherwise.

  params = (min_offset_height, max_offset_height, target_height,
            min_offset_width, max_offset_width, target_width)
  (reference :45-73).  Images are NHWC.
  """
  (min_oh, max_oh, th, min_ow, max_ow, tw) = params
  if mode == TRAIN:
    oh = int(torch.randint(min_oh, max(max_oh, min_oh + 1), (1,),
                           generator=generator).item())
    ow = int(torch.randint(min_ow, max(max_ow, min_ow + 1), (1,),
                           generator=generator).item())
  else:
    oh = (min_oh + max_oh) // 2
    ow = (min_ow + max_ow) // 2
  cropped = [img[:, oh:oh + th, ow:ow + tw, :] for img in images]
  return cropped, oh, ow


@gin.configurable
class Grasp2VecPreprocessor(
    abstract_preprocessor.SpecTransformationPreprocessor):
  """Crop, convert, random-flip (reference :76-134)."""

  _IMAGE_KEYS = ("pregrasp_image", "postgrasp_image", "goal_image")

  def __init__(self, scene_crop=(0, 40, 472, 0, 168, 472),
               goal_crop=(0, 40, 472, 0, 168, 472), **kwargs):
    self._scene_crop = scene_crop
    self._goal_crop = goal_crop
    super().__init__(**kwargs)

  def _transform_in_feature_specification(self, flat_spec, mode):
    out = tsu.TensorSpecStruct()
    for key, spec in flat_spec.items():
      if key in self._IMAGE_KEYS:
        out[key] = TSPEC(RAW_SHAPE, torch.uint8, name=spec.name,
                         data_format="jpeg",
                         dataset_key=spec.dataset_key)
      else:
        out[key] = spec
    return out

  def _preprocess_fn(self, features, labels, mode):
    scene, _, _ = maybe_crop_images(
        [features["pregrasp_image"], features["postgrasp_image"]],
        self._scene_crop, mode)
    features["pregrasp_image"], features["postgrasp_image"] = scene
    features["goal_image"] = maybe_crop_images(
        [features["goal_image"]], self._goal_crop, mode)[0][0]
    for name in self._IMAGE_KEYS:
      image = features[name]
      if image.dtype == torch.uint8:
        image = image.to(torch.float32) / 255.0
      if mode == TRAIN:
        if torch.rand(()) < 0.5:
          image = torch.flip(image, dims=[2])  # left-right (NHWC)
        if torch.rand(()) < 0.5:
          image = torch.flip(image, dims=[1])  # up-down
      features[name] = image
    return features, labels


class Embedding(nn.Module):
  """ResNet-50 spatial features -> relu -> mean pool (networks.py:24-42)."""

  def __init__(self, resnet_size: int = 50):
    super().__init__()
    self.resnet = resnet_mod.ResNet(resnet_size=resnet_size, num_classes=0)
    self.out_dim = self.resnet.out_channels

  def forward(self, image: torch.Tensor
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    spatial = resnet_mod.get_resnet50_spatial(image, self.resnet)
    spatial = F.relu(spatial)
    return spatial.mean(dim=(2, 3)), spatial


@gin.configurable
class Grasp2VecModel(abstract_model.AbstractT2RModel):
  """Scene/goal embedding towers + metric loss (reference :136-240)."""

  def __init__(self, scene_size=(472, 472), goal_size=(472, 472),
               embedding_loss_fn=losses.NPairsLoss, resnet_size: int = 50,
               **kwargs):
    super().__init__(**kwargs)
    self._scene_size = tuple(scene_size)
    self._goal_size = tuple(goal_size)
    self._embedding_loss_fn = embedding_loss_fn
    self._resnet_size = resnet_size

  @property
  def default_preprocessor_cls(self):
    return Grasp2VecPreprocessor

  def get_feature_specification(self, mode):
    spec = tsu.TensorSpecStruct()
    spec["pregrasp_image"] = TSPEC(self._scene_size + (3,), torch.float32,
                                   name="image", data_format="jpeg")
    spec["postgrasp_image"] = TSPEC(self._scene_size + (3,), torch.float32,
                                    name="postgrasp_image",
                                    data_format="jpeg")
    spec["goal_image"] = TSPEC(self._goal_size + (3,), torch.float32,
                               name="present_image", data_format="jpeg")
    return spec

  def get_label_specification(self, mode):
    return tsu.TensorSpecStruct()  # unsupervised

  def create_network(self):
    return nn.ModuleDict({
        "scene": Embedding(self._resnet_size),
        "goal": Embedding(self._resnet_size),
    })

  @staticmethod
  def _to_nchw(image: torch.Tensor) -> torch.Tensor:
    if image.dim() == 4 and image.shape[-1] == 3:
      return image.permute(0, 3, 1, 2).contiguous()
    return image

  def inference_network_fn(self, features, labels, mode, params=None):
    # One tower pass over both scene images (reference :188-190).
    scene_images = torch.cat(
        [self._to_nchw(features["pregrasp_image"]),
         self._to_nchw(features["postgrasp_image"])], dim=0)
    v, s = self.network["scene"](scene_images)
    pre_v, post_v = torch.chunk(v, 2, dim=0)
    pre_s, post_s = torch.chunk(s, 2, dim=0)
    goal_v, goal_s = self.network["goal"](
        self._to_nchw(features["goal_image"]))
    return {
        "pre_vector": pre_v,
        "post_vector": post_v,
        "pre_spatial": pre_s,
        "post_spatial": post_s,
        "goal_vector": goal_v,
        "goal_spatial": goal_s,
    }

  def model_train_fn(self, features, labels, inference_outputs, mode,
                     params=None):
    embed_loss = self._embedding_loss_fn(
        inference_outputs["pre_vector"],
        inference_outputs["goal_vector"],
        inference_outputs["post_vector"])
    if isinstance(embed_loss, tuple):  # TripletLoss returns extras
      embed_loss = embed_loss[0]
    return embed_loss, {"embed_loss": embed_loss}

  def model_eval_fn(self, features, labels, inference_outputs, train_loss,
                    train_outputs, mode, params=None):
    """Localization response statistics (reference visualization path)."""
    max_heat, max_soft = losses.get_softmax_response(
        inference_outputs["goal_vector"],
        inference_outputs["pre_spatial"])
    return {
        "embed_loss": train_loss,
        "mean_max_heat": max_heat.mean(),
        "mean_max_softmax": max_soft.mean(),
    }
