"""Grasp2Vec embedding networks (reference `research/grasp2vec/networks.py`).

The Embedding tower (ResNet-50 spatial features -> relu -> mean pool,
reference :24-42) lives in grasp2vec_model; re-exported here for module
parity, alongside the spatial-feature helper from layers/resnet
(reference `research/grasp2vec/resnet.py:537` get_resnet50_spatial).
"""

from tensor2robot_amd.layers.resnet import get_resnet50_spatial
from tensor2robot_amd.research.grasp2vec.grasp2vec_model import Embedding

__all__ = ["Embedding", "get_resnet50_spatial", "get_resnet_model"]


def get_resnet_model(resnet_size: int = 50, num_classes: int = 1001):
  """Truncated ResNet factory (reference `grasp2vec/resnet.py:505-535`):
  the grasp2vec tower cuts the last block group (block_sizes [3,4,6],
  strides [1,2,2]) to keep spatial resolution.  Returns the torch
  module; feed through get_resnet50_spatial for the pre-pool map."""
  from tensor2robot_amd.layers import resnet as resnet_lib
  return resnet_lib.ResNet(resnet_size=resnet_size,
                           num_classes=num_classes, version=2)
