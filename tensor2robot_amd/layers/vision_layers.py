"""Berkeley-Net vision torsos and pose heads.

Reference `layers/vision_layers.py`: BuildImagesToFeaturesModel :30
(conv blocks, 32 ch each, VALID padding, stride 2 on the first two
blocks, layer-norm, optional FiLM (1+gamma)x+beta pre-ReLU, final 1x1
conv then spatial softmax), BuildFILMParams :162 (plain linear),
BuildImagesToFeaturesModelHighRes :185 (multi-resolution feature sum,
PI-GPS), BuildImageFeaturesToPoseModel :277 (feature points + aux +
bias-transform -> FC stack -> pose, optional aux head).

Torch-native NCHW modules.  Normalization is LayerNorm over channels
(slim.layer_norm normalizes over all but the batch dim; we use the
channel-wise GroupNorm(1, C) equivalent which is shape-agnostic).
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin
from tensor2robot_amd.layers import spatial_softmax as ss_mod

_BLOCK_CHANNELS = 32  # reference :91 num_channels_per_block


def _norm(channels: int, norm: str) -> Optional[nn.Module]:
  if norm == "layer":
    # slim.layer_norm over [H, W, C] per sample == GroupNorm(1, C).
    return nn.GroupNorm(1, channels)
  if norm == "batch":
    return nn.BatchNorm2d(channels, eps=1e-4, momentum=0.01)
  if norm == "none":
    return None
  raise ValueError(f"unknown norm {norm}")


@gin.configurable
class ImagesToFeaturesNet(nn.Module):
  """Berkeley-Net torso (reference BuildImagesToFeaturesModel :30).

  forward(images, film_params=None) -> (features, extra) where features
  is [N, 2*num_output_maps] feature points when use_spatial_softmax else
  the [N, num_output_maps, H, W] map; extra holds the softmax map.
  film_params: [N, 2*num_blocks*32] (gammas then betas, reference
  :117-119) produced by FiLMParams.
  """

  def __init__(self, in_channels: int = 3, filter_size: int = 3,
               num_blocks: int = 5, num_output_maps: int = 32,
               normalizer: str = "layer",
               use_spatial_softmax: bool = True):
    super().__init__()
    self.num_blocks = num_blocks
    self.use_spatial_softmax = use_spatial_softmax
    convs, norms = [], []
    ch = in_channels
    for i in range(num_blocks):
      stride = 2 if i < 2 else 1  # reference :131-134
      convs.append(nn.Conv2d(ch, _BLOCK_CHANNELS, filter_size,
                             stride=stride, padding=0))  # VALID
      norms.append(_norm(_BLOCK_CHANNELS, normalizer))
      ch = _BLOCK_CHANNELS
    self.convs = nn.ModuleList(convs)
    self.norms = nn.ModuleList(norms)
    self.final_conv = nn.Conv2d(ch, num_output_maps, 1)
    self.final_norm = _norm(num_output_maps, normalizer)
    self.spatial_softmax = ss_mod.SpatialSoftmax()
    for conv in list(self.convs) + [self.final_conv]:
      nn.init.constant_(conv.bias, 0.01)  # reference biases_initializer

  def forward(self, images: torch.Tensor,
              film_params: Optional[torch.Tensor] = None
              ) -> Tuple[torch.Tensor, Dict[str, torch.Tensor]]:
    gammas = betas = None
    if film_params is not None:
      expected = 2 * self.num_blocks * _BLOCK_CHANNELS
      if film_params.shape[-1] != expected:
        raise ValueError(
            f"FiLM params last dim {film_params.shape[-1]} != {expected}")
      halves = film_params.reshape(film_params.shape[0], 2,
                                   self.num_blocks, _BLOCK_CHANNELS)
      gammas = 1.0 + halves[:, 0]  # reference :118-119
      betas = halves[:, 1]
    net = images
    for i, (conv, norm) in enumerate(zip(self.convs, self.norms)):
      net = conv(net)
      if norm is not None:
        net = norm(net)
      if gammas is not None:
        g = gammas[:, i].reshape(-1, _BLOCK_CHANNELS, 1, 1)
        b = betas[:, i].reshape(-1, _BLOCK_CHANNELS, 1, 1)
        net = g * net + b
      net = F.relu(net)
    net = self.final_conv(net)
    if self.final_norm is not None:
      net = self.final_norm(net)
    net = F.relu(net)
    if self.use_spatial_softmax:
      points, softmax = self.spatial_softmax(net)
      return points, {"softmax": softmax}
    return net, {}


@gin.configurable
class FiLMParams(nn.Module):
  """Embedding -> FiLM params, plain linear (reference BuildFILMParams :162)."""

  def __init__(self, embedding_dim: int,
               film_output_size: int = 2 * 5 * _BLOCK_CHANNELS):
    super().__init__()
    self.fc = nn.Linear(embedding_dim, film_output_size)

  def forward(self, embedding: torch.Tensor) -> torch.Tensor:
    return self.fc(embedding)


@gin.configurable
class ImagesToFeaturesNetHighRes(nn.Module):
  """Multi-resolution torso (reference :185-273, PI-GPS).

  avg_pool/2 -> conv16/2 -> conv32 -> [1x1] collected; then per block
  maxpool/2 -> conv32 -> [1x1] collected; nearest-upsample all to the
  first block's resolution, sum, final 1x1, spatial softmax.
  """

  def __init__(self, in_channels: int = 3, filter_size: int = 3,
               num_blocks: int = 5, num_output_maps: int = 32,
               normalizer: str = "batch"):
    super().__init__()
    self.num_blocks = num_blocks
    self.conv1 = nn.Conv2d(in_channels, 16, filter_size, stride=2)
    self.norm1 = _norm(16, normalizer)
    self.conv2 = nn.Conv2d(16, 32, filter_size)
    self.norm2 = _norm(32, normalizer)
    self.block_1x1 = nn.ModuleList(
        [nn.Conv2d(32, 32, 1) for _ in range(num_blocks)])
    self.block_convs = nn.ModuleList(
        [nn.Conv2d(32, 32, filter_size) for _ in range(num_blocks - 1)])
    self.block_norms = nn.ModuleList(
        [_norm(32, normalizer) for _ in range(num_blocks - 1)])
    self.final_conv = nn.Conv2d(32, num_output_maps, 1)
    self.spatial_softmax = ss_mod.SpatialSoftmax()

  def forward(self, images: torch.Tensor
              ) -> Tuple[torch.Tensor, Dict[str, torch.Tensor]]:
    net = F.avg_pool2d(images, 2, stride=2)
    net = self.conv1(net)
    if self.norm1 is not None:
      net = self.norm1(net)
    net = F.relu(net)
    net = self.conv2(net)
    if self.norm2 is not None:
      net = self.norm2(net)
    net = F.relu(net)
    block_outs = [F.relu(self.block_1x1[0](net))]
    for i in range(self.num_blocks - 1):
      net = F.max_pool2d(net, 2, stride=2, ceil_mode=False)
      net = self.block_convs[i](net)
      if self.block_norms[i] is not None:
        net = self.block_norms[i](net)
      net = F.relu(net)
      block_outs.append(F.relu(self.block_1x1[i + 1](net)))
    target = block_outs[0].shape[-2:]
    summed = sum(
        F.interpolate(b, size=target, mode="nearest") for b in block_outs)
    net = F.relu(self.final_conv(summed))
    points, softmax = self.spatial_softmax(net)
    return points, {"softmax": softmax}


@gin.configurable
class ImageFeaturesToPoseNet(nn.Module):
  """Feature points (+aux, +bias transform) -> pose.

  Reference BuildImageFeaturesToPoseModel :277-350: concat aux input,
  concat a learned `bias_transform` vector (a free parameter broadcast
  across the batch, :318-321 — a MAML trick giving the inner loop a
  direct knob), num_layers x FC(hidden_dim) with layer norm, linear
  output head, optional aux head off the raw feature points.
  """

  def __init__(self, feature_dim: int, num_outputs: int,
               aux_input_dim: int = 0, aux_output_dim: int = 0,
               hidden_dim: int = 100, num_layers: int = 2,
               bias_transform_size: int = 10):
    super().__init__()
    self.aux_output_dim = aux_output_dim
    self.bias_transform = nn.Parameter(
        torch.full((bias_transform_size,), 0.01)) \
        if bias_transform_size > 0 else None
    in_dim = feature_dim + aux_input_dim + bias_transform_size
    fcs, lns = [], []
    for _ in range(num_layers):
      fcs.append(nn.Linear(in_dim, hidden_dim))
      lns.append(nn.LayerNorm(hidden_dim))
      in_dim = hidden_dim
    self.fcs = nn.ModuleList(fcs)
    self.lns = nn.ModuleList(lns)
    self.head = nn.Linear(in_dim, num_outputs) if num_outputs else None
    self.aux_head = nn.Linear(feature_dim, aux_output_dim) \
        if aux_output_dim > 0 else None
    for fc in list(self.fcs) + ([self.head] if self.head else []):
      nn.init.normal_(fc.weight, std=0.01)
      nn.init.constant_(fc.bias, 0.01)

  def forward(self, feature_points: torch.Tensor,
              aux_input: Optional[torch.Tensor] = None
              ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    net = feature_points
    if aux_input is not None:
      net = torch.cat([net, aux_input], dim=1)
    if self.bias_transform is not None:
      bt = self.bias_transform.expand(net.shape[0], -1)
      net = torch.cat([net, bt], dim=1)
    for fc, ln in zip(self.fcs, self.lns):
      net = F.relu(ln(fc(net)))
    if self.head is not None:
      net = self.head(net)
    aux_out = self.aux_head(feature_points) if self.aux_head is not None \
        else None
    return net, aux_out
