"""Context-merge modules for convnet grasping models.

Reference `research/dql_grasping_lib/tf_modules.py`: argscope :25 (conv
defaults for grasping nets), tile_to_match_context :45, add_context :70
(broadcast-add an action/context vector across a conv feature map's
spatial extent — the CEM megabatch merge used by Grasping44).

NCHW torch-native.
"""

from __future__ import annotations

import torch



def tile_to_match_context(net: torch.Tensor,
                          context: torch.Tensor) -> torch.Tensor:
  """[B, ...] -> [B, M, ...] where M = context.shape[1] (reference :45)."""
  num_samples = context.shape[1]
  net = net.unsqueeze(1)
  return net.expand(net.shape[0], num_samples, *net.shape[2:])


def add_context(net: torch.Tensor, context: torch.Tensor) -> torch.Tensor:
  """Broadcast-add context to conv features (reference :70-93).

  net: [B, C, H, W]; context: [B * M, C] (M action samples per image).
  Returns [B * M, C, H, W].
  """
  b, c, h, w = net.shape
  if context.shape[-1] != c:
    raise ValueError(
        f"context dim {context.shape[-1]} != net channels {c}")
  context = context.reshape(b, -1, c)
  tiled = tile_to_match_context(net, context)       # [B, M, C, H, W]
  net = tiled.reshape(-1, c, h, w)
  return net + context.reshape(-1, c, 1, 1)
