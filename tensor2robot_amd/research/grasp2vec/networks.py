"""Grasp2Vec embedding networks (reference `research/grasp2vec/networks.py`).

The Embedding tower (ResNet-50 spatial features -> relu -> mean pool,
reference :24-42) lives in grasp2vec_model; re-exported here for module
parity, alongside the spatial-feature helper from layers/resnet
(reference `research/grasp2vec/resnet.py:537` get_resnet50_spatial).
"""

from tensor2robot_amd.layers.resnet import get_resnet50_spatial
from tensor2robot_amd.research.grasp2vec.grasp2vec_model import Embedding

__all__ = ["Embedding", "get_resnet50_spatial"]
