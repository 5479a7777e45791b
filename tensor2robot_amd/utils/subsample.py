"""Temporal subsampling of padded episode sequences.

Reference `utils/subsample.py:22-244`: index generators over per-example
sequence lengths — uniform (consistent frame rate, always includes the
last frame), random without first/last, random with endpoints (without
replacement when long enough, else with), randomized-boundary window
variant, and a numpy twin of the endpoint sampler.
All return int64 index tensors [B, min_length], sorted per row.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from tensor2robot_amd import gin


def _per_row(fn, sequence_lengths: torch.Tensor,
             min_length: int) -> torch.Tensor:
  rows = [fn(int(l)) for l in sequence_lengths.reshape(-1)]
  return torch.stack(rows).to(torch.int64)


def get_uniform_subsample_indices(sequence_lengths: torch.Tensor,
                                  min_length: int) -> torch.Tensor:
  """Consistent frame-rate indices, last frame guaranteed (reference :22)."""

  def one(sequence_length: int) -> torch.Tensor:
    idx = torch.arange(min_length, dtype=torch.float64)
    idx = torch.round(idx * (sequence_length - 1) / min_length)
    idx = (sequence_length - 1) - idx
    return torch.sort(idx.to(torch.int64)).values

  return _per_row(one, sequence_lengths, min_length)


def get_subsample_indices_nofirstlast(
    sequence_lengths: torch.Tensor, min_length: int,
    generator: Optional[torch.Generator] = None) -> torch.Tensor:
  """Random with replacement, no endpoint guarantee (reference :53)."""

  def one(sequence_length: int) -> torch.Tensor:
    idx = torch.floor(torch.rand(min_length, generator=generator)
                      * sequence_length).to(torch.int64)
    return torch.sort(idx).values

  return _per_row(one, sequence_lengths, min_length)


def get_subsample_indices(sequence_lengths: torch.Tensor,
                          min_length: int,
                          generator: Optional[torch.Generator] = None
                          ) -> torch.Tensor:
  """Random indices incl. first+last frames (reference :82-138)."""

  def one(sequence_length: int) -> torch.Tensor:
    if min_length == 1:
      return torch.floor(torch.rand(1, generator=generator)
                         * sequence_length).to(torch.int64)
    if sequence_length >= min_length:
      perm = torch.randperm(max(0, sequence_length - 2),
                            generator=generator) + 1
      middle = perm[:min_length - 2]
    else:
      middle = torch.floor(torch.rand(min_length - 2,
                                      generator=generator)
                           * sequence_length).to(torch.int64)
    full = torch.cat([torch.tensor([0]), middle.to(torch.int64),
                      torch.tensor([sequence_length - 1])])
    return torch.sort(full).values

  return _per_row(one, sequence_lengths, min_length)


@gin.configurable
def get_subsample_indices_randomized_boundary(
    sequence_lengths: torch.Tensor, min_length: int, min_delta_t: int,
    max_delta_t: int, generator: Optional[torch.Generator] = None
    ) -> torch.Tensor:
  """Endpoint sampling within a random time window (reference :141-216)."""

  def one(sequence_length: int) -> torch.Tensor:
    delta = int(torch.randint(min_delta_t, max_delta_t + 1, (1,),
                              generator=generator).item())
    delta = min(sequence_length, delta)
    start = int(torch.randint(0, sequence_length - delta + 1, (1,),
                              generator=generator).item())
    end = start + delta - 1
    if min_length == 1:
      return torch.randint(start, max(start + 1, end), (1,),
                           generator=generator)
    if delta >= min_length:
      perm = torch.randperm(max(0, end - start - 1),
                            generator=generator) + start + 1
      middle = perm[:min_length - 2]
    else:
      middle = start + torch.floor(
          torch.rand(min_length - 2, generator=generator)
          * delta).to(torch.int64)
    full = torch.cat([torch.tensor([start]), middle.to(torch.int64),
                      torch.tensor([end])])
    return torch.sort(full).values

  return _per_row(one, sequence_lengths, min_length)


def get_np_subsample_indices(sequence_lengths: np.ndarray,
                             min_length: int) -> np.ndarray:
  """Numpy twin of get_subsample_indices (reference :220-244)."""

  def one(sequence_length: int) -> np.ndarray:
    if min_length == 1:
      return np.random.randint(0, sequence_length, size=(1,))
    if sequence_length >= min_length:
      arr = np.arange(1, sequence_length - 1)
      np.random.shuffle(arr)
      middle = arr[:min_length - 2]
    else:
      middle = np.random.randint(0, sequence_length,
                                 size=[min_length - 2])
    return np.sort(np.concatenate([[0], middle, [sequence_length - 1]]))

  batch_size = np.asarray(sequence_lengths).shape[0]
  out = np.zeros((batch_size, min_length), np.int64)
  for i in range(batch_size):
    out[i] = one(int(sequence_lengths[i]))
  return out


def subsample_sequence(sequence: torch.Tensor,
                       indices: torch.Tensor) -> torch.Tensor:
  """Gathers [B, T, ...] at per-row indices [B, K] -> [B, K, ...]."""
  b, k = indices.shape
  idx = indices.reshape(b, k, *([1] * (sequence.dim() - 2)))
  idx = idx.expand(b, k, *sequence.shape[2:])
  return torch.gather(sequence, 1, idx)
