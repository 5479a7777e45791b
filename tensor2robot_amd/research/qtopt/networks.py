"""QT-Opt Grasping44 critic network (MI355X-native).

Architecture parity with the reference's
`research/qtopt/networks.py:299-580` Grasping44FlexibleGraspParams
(num_convs=[6,6,3], hid_layers=2, 64 channels):

  472x472x3 -> conv 6x6/2 (64, BN+ReLU) -> maxpool 3x3/3
  -> 6x conv 5x5 SAME (64, BN+ReLU) -> maxpool 3x3/3            [79 -> 27]
  action -> FC 256 (BN+ReLU) -> FC 64 (BN+ReLU) -> broadcast-add as
  spatial context (CEM megabatch: image embedding tiled
  action_batch_size times, reference :515-521)
  -> 6x conv 3x3 SAME -> maxpool 2x2/2 -> 3x conv 3x3 VALID     [27->14->8]
  -> flatten -> 2x FC 64 (BN+ReLU) -> FC num_classes -> sigmoid

Pooling uses ceil_mode to reproduce TF SAME-pool geometry.  The module
runs in channels_last memory format on the HIP device; BN+ReLU pairs and
the sigmoid+log-loss epilogue are fused HIP kernels (tensor2robot_amd/ops)
on the hot path.
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import nn

from tensor2robot_amd import gin
from tensor2robot_amd.ops import fused_bn
from tensor2robot_amd.ops import conv as mfma_conv
from tensor2robot_amd.ops import maxpool as fused_maxpool


def _conv_bn_relu(in_ch: int, out_ch: int, kernel: int, stride: int = 1,
                  padding: int = 0) -> nn.Sequential:
  # MFMAConv2d self-dispatches: the hand-written kernel on supported
  # stride-1 shapes (3x3 blocks, 3.3-4x vs MIOpen), F.conv2d otherwise.
  return nn.Sequential(
      mfma_conv.MFMAConv2d(in_ch, out_ch, kernel, stride=stride,
                           padding=padding, bias=False),
      fused_bn.FusedBatchNormReLU(out_ch, eps=0.001, momentum=0.003),
  )


@gin.configurable
class Grasping44(nn.Module):
  """The QT-Opt grasping critic Q(image, action) -> [0, 1]."""

  def __init__(self, action_dim: int = 10, channels: int = 64,
               num_convs: Tuple[int, int, int] = (6, 6, 3),
               hid_layers: int = 2, num_classes: int = 1):
    super().__init__()
    self.action_dim = action_dim
    self.num_classes = num_classes
    ch = channels
    self.conv1 = mfma_conv.MFMAConv2d(3, ch, 6, stride=2, padding=2,
                                      bias=False)
    self.bn1 = fused_bn.FusedBatchNormReLU(ch, eps=0.001, momentum=0.003)
    self.pool1 = fused_maxpool.FusedMaxPool2d(3, ceil_mode=True)
    self.block1 = nn.Sequential(*[
        _conv_bn_relu(ch, ch, 5, padding=2) for _ in range(num_convs[0])])
    self.pool2 = fused_maxpool.FusedMaxPool2d(3, ceil_mode=True)
    self.fc_action1 = nn.Linear(action_dim, 256)
    self.bn_action = fused_bn.FusedBatchNormReLU(256, eps=0.001,
                                                 momentum=0.003)
    self.fc_action2 = nn.Linear(256, ch, bias=False)
    self.bn_action2 = fused_bn.FusedBatchNormReLU(ch, eps=0.001,
                                                  momentum=0.003)
    self.block2 = nn.Sequential(*[
        _conv_bn_relu(ch, ch, 3, padding=1) for _ in range(num_convs[1])])
    self.pool3 = fused_maxpool.FusedMaxPool2d(2, ceil_mode=True)
    self.block3 = nn.Sequential(*[
        _conv_bn_relu(ch, ch, 3, padding=0) for _ in range(num_convs[2])])
    # 472 -> 236 -> 79 -> 27 -> 14 -> 12 -> 10 -> 8 spatial.
    self.fc_head = nn.ModuleList()
    in_dim = ch * 8 * 8
    for _ in range(hid_layers):
      self.fc_head.append(nn.Sequential(
          nn.Linear(in_dim, 64, bias=False),
          fused_bn.FusedBatchNormReLU(64, eps=0.001, momentum=0.003)))
      in_dim = 64
    self.logit = nn.Linear(64, num_classes)

  def embed_image(self, image: torch.Tensor) -> torch.Tensor:
    """Image tower up to the action-merge point (pool2 output)."""
    net = self.conv1(image)
    net = self.bn1(net)
    net = self.pool1(net)
    net = self.block1(net)
    return self.pool2(net)

  def embed_action(self, action: torch.Tensor) -> torch.Tensor:
    a = self.fc_action1(action)
    a = self.bn_action(a)
    a = self.fc_action2(a)
    a = self.bn_action2(a)
    return a

  def head(self, net: torch.Tensor) -> torch.Tensor:
    net = self.block2(net)
    net = self.pool3(net)
    net = self.block3(net)
    net = net.flatten(1)
    for fc in self.fc_head:
      net = fc(net)
    return self.logit(net)

  def forward(self, image: torch.Tensor, action: torch.Tensor
              ) -> torch.Tensor:
    """image: [N,3,H,W]; action: [N,A] or [N,S,A] (CEM megabatch).

    Returns Q LOGITS of shape [N] (or [N,S] for the megabatch); apply
    sigmoid for probabilities (kept out of the module so training can use
    the numerically-stable fused logits loss).
    """
    tile_batch = action.dim() == 3
    action_samples = action.shape[1] if tile_batch else 1
    if tile_batch:
      action = action.reshape(-1, action.shape[-1])  # megabatch collapse
    emb = self.embed_image(image)
    if tile_batch:
      # Tile image embedding across the action samples (reference :515-521).
      emb = emb.repeat_interleave(action_samples, dim=0)
    context = self.embed_action(action)
    net = emb + context[:, :, None, None]
    logits = self.head(net)
    if tile_batch:
      if self.num_classes > 1:
        return logits.reshape(-1, action_samples, self.num_classes)
      return logits.reshape(-1, action_samples)
    if self.num_classes == 1:
      return logits.squeeze(-1)
    return logits
