"""Grasp2Vec heatmap / localization visualization helpers.

Reference `research/grasp2vec/visualization.py`: add_heatmap_summary
(goal embedding dotted over scene spatial features -> normalized
heatmap image) and add_spatial_softmax (soft arg-max of the heatmap
overlaid on the scene).  Torch-native: these return tensors; the caller
logs them through the summary writer.
"""

from __future__ import annotations

from typing import Tuple

import torch

from tensor2robot_amd.layers import spatial_softmax as ss_mod


def compute_heatmap(goal_vector: torch.Tensor,
                    scene_spatial: torch.Tensor) -> torch.Tensor:
  """[N, D] x [N, D, H, W] -> [N, 1, H, W] min-max-normalized heatmap."""
  b, d = goal_vector.shape
  heat = (scene_spatial * goal_vector.reshape(b, d, 1, 1)).sum(
      dim=1, keepdim=True)
  flat = heat.flatten(1)
  lo = flat.min(dim=1).values.reshape(b, 1, 1, 1)
  hi = flat.max(dim=1).values.reshape(b, 1, 1, 1)
  return (heat - lo) / (hi - lo + 1e-12)


def heatmap_keypoints(heatmaps: torch.Tensor
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
  """Soft arg-max keypoints of [N, 1, H, W] heatmaps -> ([N, 2], map)."""
  return ss_mod.SpatialSoftmax()(heatmaps)
