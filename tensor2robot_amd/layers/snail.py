"""SNAIL temporal-convolution + causal-attention blocks.

Reference `layers/snail.py`: CausalConv :29 (left-padded dilated conv1d),
DenseBlock :54 (tanh x sigmoid gate, channel concat), TCBlock :72
(dilations 2^1..2^ceil(log2 T)), CausallyMaskedSoftmax :89 (upper-tri
-inf mask), AttentionBlock :113 (single-head causal KV attention,
concatenated onto the input).

All modules take sequences as [N, T, D] (episode time-series, SURVEY
§5.7 — T is small, ~40, so the O(T^2) attention is cheap).
"""

from __future__ import annotations

import math

import torch
from torch import nn
import torch.nn.functional as F

from tensor2robot_amd import gin


class CausalConv(nn.Module):
  """Left-padded dilated 1-D conv (reference :29-51)."""

  def __init__(self, in_dim: int, out_dim: int, kernel_size: int = 2,
               dilation: int = 1):
    super().__init__()
    self.pad = (kernel_size - 1) * dilation
    self.conv = nn.Conv1d(in_dim, out_dim, kernel_size, dilation=dilation)

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    # [N, T, D] -> [N, D, T], left pad so output t sees inputs <= t.
    y = F.pad(x.transpose(1, 2), (self.pad, 0))
    return self.conv(y).transpose(1, 2)


class DenseBlock(nn.Module):
  """Gated causal conv, output concatenated to input (reference :54-69)."""

  def __init__(self, in_dim: int, filters: int, dilation: int = 1):
    super().__init__()
    self.conv_f = CausalConv(in_dim, filters, dilation=dilation)
    self.conv_g = CausalConv(in_dim, filters, dilation=dilation)
    self.out_dim = in_dim + filters

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    activations = torch.tanh(self.conv_f(x)) * torch.sigmoid(self.conv_g(x))
    return torch.cat([x, activations], dim=-1)


class TCBlock(nn.Module):
  """Stack of DenseBlocks with dilations 2^0..2^(ceil(log2 T)-1).

  Reference :72-86 (dilation doubling until the receptive field covers
  the sequence length).
  """

  def __init__(self, in_dim: int, sequence_length: int, filters: int):
    super().__init__()
    num_layers = max(1, int(math.ceil(math.log2(max(2, sequence_length)))))
    blocks = []
    dim = in_dim
    for i in range(num_layers):
      block = DenseBlock(dim, filters, dilation=2 ** i)
      dim = block.out_dim
      blocks.append(block)
    self.blocks = nn.ModuleList(blocks)
    self.out_dim = dim

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    for block in self.blocks:
      x = block(x)
    return x


def causally_masked_softmax(logits: torch.Tensor) -> torch.Tensor:
  """Softmax over the last axis with future positions masked.

  Reference :89-110: strictly-upper-triangular -inf mask so position t
  attends to positions <= t.  logits [..., T, T].
  """
  t = logits.shape[-1]
  mask = torch.triu(torch.ones(t, t, dtype=torch.bool,
                               device=logits.device), diagonal=1)
  return torch.softmax(logits.masked_fill(mask, float("-inf")), dim=-1)


class AttentionBlock(nn.Module):
  """Single-head causal KV attention, output concat (reference :113-134)."""

  def __init__(self, in_dim: int, key_size: int, value_size: int):
    super().__init__()
    self.key = nn.Linear(in_dim, key_size)
    self.query = nn.Linear(in_dim, key_size)
    self.value = nn.Linear(in_dim, value_size)
    self.scale = 1.0 / math.sqrt(key_size)
    self.out_dim = in_dim + value_size

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    q, k, v = self.query(x), self.key(x), self.value(x)
    logits = torch.matmul(q, k.transpose(-1, -2)) * self.scale
    attn = causally_masked_softmax(logits)
    read = torch.matmul(attn, v)
    return torch.cat([x, read], dim=-1)


@gin.configurable
class SNAILNet(nn.Module):
  """TC + attention policy torso (reference `bcz_networks.py:81` SNAIL)."""

  def __init__(self, in_dim: int, sequence_length: int,
               filters: int = 32, key_size: int = 32, value_size: int = 32,
               out_dim: int = 64):
    super().__init__()
    self.attn1 = AttentionBlock(in_dim, key_size, value_size)
    self.tc1 = TCBlock(self.attn1.out_dim, sequence_length, filters)
    self.attn2 = AttentionBlock(self.tc1.out_dim, key_size, value_size)
    self.tc2 = TCBlock(self.attn2.out_dim, sequence_length, filters)
    self.head = nn.Linear(self.tc2.out_dim, out_dim)
    self.out_dim = out_dim

  def forward(self, x: torch.Tensor) -> torch.Tensor:
    return self.head(self.tc2(self.attn2(self.tc1(self.attn1(x)))))
