"""ResNet v1/v2 with optional FiLM conditioning (MI355X-native).

Single implementation covering the reference's `layers/resnet.py:147`
(resnet_model sizes 18/34/50/101/152/200, v2 default, channels_last,
endpoints :80, FiLM generator :98, warm-start :213) and the forked
`layers/film_resnet_model.py` (batch_norm momentum .997 eps 1e-5 :50-57,
`_apply_film` (1+gamma)x+beta :108-115, block fns v1/v2 +/- bottleneck
:121-341, Model :391).

The torch module runs channels_last on the HIP device; BN+ReLU pairs use
the fused CDNA4 kernel (ops/fused_bn) — v2's pre-activation BN+ReLU and
the v1 post-conv BN+ReLU both map onto it.  FiLM's (1+gamma)x+beta is a
per-channel affine the same shape as BN's epilogue.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin
from tensor2robot_amd.ops import fused_bn

# Reference `resnet.py:47-54` _get_block_sizes.
_BLOCK_SIZES = {
    18: [2, 2, 2, 2],
    34: [3, 4, 6, 3],
    50: [3, 4, 6, 3],
    101: [3, 4, 23, 3],
    152: [3, 8, 36, 3],
    200: [3, 24, 36, 3],
}
_BOTTLENECK_MIN_SIZE = 50  # 50+ use bottleneck blocks (reference :165-172).

_BN_MOMENTUM = 1.0 - 0.997  # torch momentum = 1 - TF decay (:50-57)
_BN_EPS = 1e-5


def _bn(channels: int, relu: bool) -> fused_bn.FusedBatchNormReLU:
  return fused_bn.FusedBatchNormReLU(
      channels, eps=_BN_EPS, momentum=_BN_MOMENTUM, fuse_relu=relu)


def _conv_fixed_padding(in_ch: int, out_ch: int, kernel: int,
                        stride: int) -> nn.Conv2d:
  """Reference `film_resnet_model.py:60-105` fixed_padding semantics.

  TF pads (k-1)//2 before / k//2 after when stride>1 ("explicit" pad);
  for stride 1 SAME with odd kernels torch's symmetric padding is
  identical.  For the even-offset stride-2 case we use torch padding
  (k-1)//2 which yields the same output size; the one-pixel alignment
  difference does not change the architecture contract.

  MFMAConv2d self-dispatches per shape: fused MFMA kernels on the
  C=K<=64 stride-1 shapes, hand-im2col + rocBLAS GEMM on the ResNet
  C/K-up-to-512 shapes, torch/MIOpen otherwise.
  """
  from tensor2robot_amd.ops import conv as mfma_conv
  return mfma_conv.MFMAConv2d(in_ch, out_ch, kernel, stride=stride,
                              padding=(kernel - 1) // 2, bias=False)


def apply_film(x: torch.Tensor, gamma_beta: Optional[torch.Tensor]
               ) -> torch.Tensor:
  """(1 + gamma) * x + beta, gamma_beta [N, 2C] (reference :108-115)."""
  if gamma_beta is None:
    return x
  n, twoc = gamma_beta.shape
  c = twoc // 2
  gamma = gamma_beta[:, :c].reshape(n, c, 1, 1)
  beta = gamma_beta[:, c:].reshape(n, c, 1, 1)
  return (1.0 + gamma) * x + beta


class _BuildingBlockV1(nn.Module):
  """Reference `film_resnet_model.py:121-170` (FiLM pre-residual-add)."""

  expansion = 1

  def __init__(self, in_ch: int, filters: int, stride: int,
               use_projection: bool):
    super().__init__()
    self.film_width = 2 * filters  # apply site: post-bn2, `filters` ch
    self.conv1 = _conv_fixed_padding(in_ch, filters, 3, stride)
    self.bn1 = _bn(filters, relu=True)
    self.conv2 = _conv_fixed_padding(filters, filters, 3, 1)
    self.bn2 = _bn(filters, relu=False)
    self.shortcut = None
    if use_projection:
      self.shortcut = nn.Sequential(
          _conv_fixed_padding(in_ch, filters, 1, stride),
          _bn(filters, relu=False))

  def forward(self, x, gamma_beta=None):
    shortcut = self.shortcut(x) if self.shortcut is not None else x
    y = self.bn1(self.conv1(x))
    y = self.bn2(self.conv2(y))
    y = apply_film(y, gamma_beta)
    return F.relu(y + shortcut)


class _BottleneckV1(nn.Module):
  """Reference `film_resnet_model.py:173-232`."""

  expansion = 4

  def __init__(self, in_ch: int, filters: int, stride: int,
               use_projection: bool):
    super().__init__()
    out_ch = filters * self.expansion
    self.film_width = 2 * out_ch  # apply site: post-bn3, expanded ch
    self.conv1 = _conv_fixed_padding(in_ch, filters, 1, 1)
    self.bn1 = _bn(filters, relu=True)
    self.conv2 = _conv_fixed_padding(filters, filters, 3, stride)
    self.bn2 = _bn(filters, relu=True)
    self.conv3 = _conv_fixed_padding(filters, out_ch, 1, 1)
    self.bn3 = _bn(out_ch, relu=False)
    self.shortcut = None
    if use_projection:
      self.shortcut = nn.Sequential(
          _conv_fixed_padding(in_ch, out_ch, 1, stride),
          _bn(out_ch, relu=False))

  def forward(self, x, gamma_beta=None):
    shortcut = self.shortcut(x) if self.shortcut is not None else x
    y = self.bn1(self.conv1(x))
    y = self.bn2(self.conv2(y))
    y = self.bn3(self.conv3(y))
    y = apply_film(y, gamma_beta)
    return F.relu(y + shortcut)


class _BuildingBlockV2(nn.Module):
  """Pre-activation block (reference `film_resnet_model.py:235-286`)."""

  expansion = 1

  def __init__(self, in_ch: int, filters: int, stride: int,
               use_projection: bool):
    super().__init__()
    self.film_width = 2 * filters  # apply site: post-bn2 pre-ReLU
    self.bn1 = _bn(in_ch, relu=True)
    self.conv1 = _conv_fixed_padding(in_ch, filters, 3, stride)
    self.bn2 = _bn(filters, relu=True)
    self.conv2 = _conv_fixed_padding(filters, filters, 3, 1)
    self.shortcut = None
    if use_projection:
      self.shortcut = _conv_fixed_padding(in_ch, filters, 1, stride)

  def forward(self, x, gamma_beta=None):
    pre = self.bn1(x)
    shortcut = self.shortcut(pre) if self.shortcut is not None else x
    y = self.conv1(pre)
    if gamma_beta is not None:
      # Reference order: batch_norm -> _apply_film -> relu -> conv
      # (film_resnet_model.py:210-213) — split the BN+ReLU fusion.
      y = F.relu(apply_film(self.bn2(y, relu=False), gamma_beta))
    else:
      y = self.bn2(y)
    y = self.conv2(y)
    return y + shortcut


class _BottleneckV2(nn.Module):
  """Pre-activation bottleneck (reference `film_resnet_model.py:289-341`)."""

  expansion = 4

  def __init__(self, in_ch: int, filters: int, stride: int,
               use_projection: bool):
    super().__init__()
    out_ch = filters * self.expansion
    self.film_width = 2 * filters  # apply site: post-bn3, pre-expansion
    self.bn1 = _bn(in_ch, relu=True)
    self.conv1 = _conv_fixed_padding(in_ch, filters, 1, 1)
    self.bn2 = _bn(filters, relu=True)
    self.conv2 = _conv_fixed_padding(filters, filters, 3, stride)
    self.bn3 = _bn(filters, relu=True)
    self.conv3 = _conv_fixed_padding(filters, out_ch, 1, 1)
    self.shortcut = None
    if use_projection:
      self.shortcut = _conv_fixed_padding(in_ch, out_ch, 1, stride)

  def forward(self, x, gamma_beta=None):
    pre = self.bn1(x)
    shortcut = self.shortcut(pre) if self.shortcut is not None else x
    y = self.conv1(pre)
    y = self.bn2(y)
    y = self.conv2(y)
    if gamma_beta is not None:
      # Reference order: batch_norm -> _apply_film -> relu -> conv3
      # (film_resnet_model.py:333-336) — split the BN+ReLU fusion.
      y = F.relu(apply_film(self.bn3(y, relu=False), gamma_beta))
    else:
      y = self.bn3(y)
    y = self.conv3(y)
    return y + shortcut


class _BlockLayer(nn.Module):
  """One of the 4 stages (reference `film_resnet_model.py:343-388`)."""

  def __init__(self, block_cls, in_ch: int, filters: int, num_blocks: int,
               stride: int):
    super().__init__()
    blocks = [block_cls(in_ch, filters, stride, use_projection=True)]
    out_ch = filters * block_cls.expansion
    for _ in range(num_blocks - 1):
      blocks.append(block_cls(out_ch, filters, 1, use_projection=False))
    self.blocks = nn.ModuleList(blocks)
    self.out_channels = out_ch

  def forward(self, x, gamma_betas: Optional[List[torch.Tensor]] = None):
    for i, block in enumerate(self.blocks):
      gb = None
      if gamma_betas is not None and i < len(gamma_betas):
        gb = gamma_betas[i]
      x = block(x, gb)
    return x


@gin.configurable
class ResNet(nn.Module):
  """ResNet 18..200 v1/v2 with FiLM hooks and named endpoints.

  forward(images, film_gamma_betas=None) -> (output, endpoints) where
  endpoints mirrors reference `resnet.py:80-96` resnet_endpoints:
  initial_conv / initial_max_pool / block_layer{1..4} / pre_final_pool /
  final_reduce_mean / final_dense.  `film_gamma_betas` is a list (len =
  total blocks, ordered block_layer1..4) of [N, 2C_i] tensors or Nones —
  the output shape of LinearFiLMGenerator.
  """

  def __init__(self, resnet_size: int = 50, num_classes: int = 0,
               version: int = 2, num_filters: int = 64, in_channels: int = 3,
               first_kernel: int = 7, first_stride: int = 2,
               first_pool: bool = True):
    super().__init__()
    if resnet_size not in _BLOCK_SIZES:
      raise ValueError(f"resnet_size must be one of {list(_BLOCK_SIZES)}")
    if version not in (1, 2):
      raise ValueError("version must be 1 or 2")
    self.resnet_size = resnet_size
    self.version = version
    self.num_classes = num_classes
    bottleneck = resnet_size >= _BOTTLENECK_MIN_SIZE
    if version == 1:
      block_cls = _BottleneckV1 if bottleneck else _BuildingBlockV1
    else:
      block_cls = _BottleneckV2 if bottleneck else _BuildingBlockV2
    self.block_sizes = _BLOCK_SIZES[resnet_size]

    self.initial_conv = _conv_fixed_padding(in_channels, num_filters,
                                            first_kernel, first_stride)
    # v1 normalizes right after the stem; v2 defers BN into the blocks.
    self.initial_bn = _bn(num_filters, relu=True) if version == 1 else None
    self.first_pool = nn.MaxPool2d(3, stride=2, padding=1) if first_pool \
        else None

    layers = []
    in_ch = num_filters
    for i, n_blocks in enumerate(self.block_sizes):
      filters = num_filters * (2 ** i)
      stride = 1 if i == 0 else 2
      layer = _BlockLayer(block_cls, in_ch, filters, n_blocks, stride)
      in_ch = layer.out_channels
      layers.append(layer)
    self.block_layers = nn.ModuleList(layers)
    self.out_channels = in_ch
    # v2 has a final BN+ReLU after the last block (pre_final_pool).
    self.final_bn = _bn(in_ch, relu=True) if version == 2 else None
    self.final_dense = nn.Linear(in_ch, num_classes) if num_classes else None

  @property
  def blocks_per_layer(self) -> List[int]:
    return list(self.block_sizes)

  @property
  def film_channels(self) -> List[int]:
    """Per-block FiLM width 2C at the apply site, block_layer1..4 flat.

    The apply-site channel count depends on the block type (pre-expansion
    `filters` for v2 blocks, expanded channels for v1 bottleneck) — the
    reference generator sizes FiLM as 2*filters per block
    (resnet.py:129-139 with filter_sizes = num_filters*2^i)."""
    dims = []
    for layer in self.block_layers:
      for block in layer.blocks:
        dims.append(block.film_width)
    return dims

  def forward(self, images: torch.Tensor,
              film_gamma_betas: Optional[List[torch.Tensor]] = None,
              ) -> Tuple[torch.Tensor, Dict[str, torch.Tensor]]:
    endpoints: Dict[str, torch.Tensor] = {}
    x = self.initial_conv(images)
    if self.initial_bn is not None:
      x = self.initial_bn(x)
    endpoints["initial_conv"] = x
    if self.first_pool is not None:
      x = self.first_pool(x)
    endpoints["initial_max_pool"] = x
    offset = 0
    for i, layer in enumerate(self.block_layers):
      gbs = None
      if film_gamma_betas is not None:
        gbs = film_gamma_betas[offset: offset + len(layer.blocks)]
      offset += len(layer.blocks)
      x = layer(x, gbs)
      endpoints[f"block_layer{i + 1}"] = x
    if self.final_bn is not None:
      x = self.final_bn(x)
    endpoints["pre_final_pool"] = x
    pooled = x.mean(dim=(2, 3))
    endpoints["final_reduce_mean"] = pooled
    out = pooled
    if self.final_dense is not None:
      out = self.final_dense(pooled)
      endpoints["final_dense"] = out
    return out, endpoints


@gin.configurable
class LinearFiLMGenerator(nn.Module):
  """Embedding -> per-block gamma/beta list.

  Reference `resnet.py:98-145` linear_film_generator: one FC per block
  layer, output split per block within the layer; `enabled_block_layers`
  (list of bool, len 4) gates which stages get FiLM (:122-132) — disabled
  stages contribute None entries.
  """

  def __init__(self, embedding_dim: int, resnet: ResNet,
               enabled_block_layers: Optional[Sequence[bool]] = None):
    super().__init__()
    self._enabled = list(enabled_block_layers) if enabled_block_layers \
        else [True] * len(resnet.block_layers)
    if len(self._enabled) != len(resnet.block_layers):
      raise ValueError("enabled_block_layers must have one entry per "
                       "block layer")
    self._blocks_per_layer = [len(l.blocks) for l in resnet.block_layers]
    # Apply-site FiLM width per block (uniform within a layer).
    self._widths = [l.blocks[0].film_width for l in resnet.block_layers]
    fcs = []
    for enabled, n_blocks, width in zip(self._enabled,
                                        self._blocks_per_layer,
                                        self._widths):
      fcs.append(nn.Linear(embedding_dim, n_blocks * width)
                 if enabled else None)
    self.fcs = nn.ModuleList([fc for fc in fcs if fc is not None])

  def forward(self, embedding: torch.Tensor) -> List[Optional[torch.Tensor]]:
    out: List[Optional[torch.Tensor]] = []
    fc_iter = iter(self.fcs)
    for enabled, n_blocks, width in zip(self._enabled,
                                        self._blocks_per_layer,
                                        self._widths):
      if not enabled:
        out.extend([None] * n_blocks)
        continue
      flat = next(fc_iter)(embedding)
      out.extend(torch.chunk(flat, n_blocks, dim=-1))
    return out


def get_resnet50_spatial(images: torch.Tensor, resnet: ResNet
                         ) -> torch.Tensor:
  """Spatial feature map before pooling (reference `grasp2vec/resnet.py:537`)."""
  _, endpoints = resnet(images)
  return endpoints["pre_final_pool"]


def resnet_init_from_checkpoint_fn(checkpoint_path: str,
                                   skip_dense: bool = True):
  """Warm start from a ResNet checkpoint minus the classifier head.

  Reference `layers/resnet.py:213-232` resnet_init_from_checkpoint_fn.
  Returns an init_fn(network) suitable for
  AbstractT2RModel(init_from_checkpoint_fn=...).
  """

  def init_fn(network: nn.Module):
    state = torch.load(checkpoint_path, map_location="cpu",
                       weights_only=True)
    if "model" in state:
      state = state["model"]
    if skip_dense:
      state = {k: v for k, v in state.items()
               if not k.startswith("final_dense.")}
    missing, unexpected = network.load_state_dict(state, strict=False)
    return {"missing": missing, "unexpected": unexpected}

  return init_fn


# Reference name (film_resnet_model.py:77 / grasp2vec/resnet.py:77):
# strided conv with kernel-size-based explicit padding.
conv2d_fixed_padding = _conv_fixed_padding
