"""Checkpoint save/restore/GC for the native train loop.

Reference semantics: TF Saver via scaffold with `max_to_keep` /
`keep_checkpoint_every_n_hours` (`models/abstract_model.py:786-804`), EMA
swapping-saver (checkpoints hold AVERAGED params, training state does not —
`models/optimizers.py:133-159`), resume restores global_step
(`utils/train_eval_test.py:204-247`), and defensive concurrency: write to a
tmp name + atomic rename, eval works on a backup copy
(`utils/train_eval.py:616-733`).
"""

from __future__ import annotations

import os
import re
import shutil
import time
from typing import Dict, List, Optional

import torch

CHECKPOINT_INDEX = "checkpoint"
_CKPT_RE = re.compile(r"model\.ckpt-(\d+)\.pt$")


def checkpoint_path(model_dir: str, step: int) -> str:
  return os.path.join(model_dir, f"model.ckpt-{step}.pt")


def list_checkpoints(model_dir: str) -> List[str]:
  if not os.path.isdir(model_dir):
    return []
  out = []
  for name in os.listdir(model_dir):
    m = _CKPT_RE.match(name)
    if m:
      out.append((int(m.group(1)), os.path.join(model_dir, name)))
  return [p for _, p in sorted(out)]


def latest_checkpoint(model_dir: str) -> Optional[str]:
  ckpts = list_checkpoints(model_dir)
  return ckpts[-1] if ckpts else None


def global_step_from_path(path: str) -> int:
  m = _CKPT_RE.search(path)
  if not m:
    raise ValueError(f"Not a checkpoint path: {path}")
  return int(m.group(1))


def wait_for_checkpoint(model_dir: str, last_seen: Optional[str] = None,
                        timeout: float = 60.0,
                        poll_interval: float = 0.5) -> Optional[str]:
  """Polls for a checkpoint newer than last_seen (checkpoints_iterator)."""
  deadline = time.time() + timeout
  while True:
    latest = latest_checkpoint(model_dir)
    if latest and latest != last_seen:
      return latest
    if time.time() >= deadline:
      return None
    time.sleep(poll_interval)


class Checkpointer:
  """Save/restore with max_to_keep GC and EMA swapping semantics."""

  def __init__(self, model_dir: str, max_to_keep: int = 5,
               keep_checkpoint_every_n_hours: Optional[float] = None):
    self._model_dir = model_dir
    self._max_to_keep = max_to_keep
    self._keep_every_s = (keep_checkpoint_every_n_hours * 3600.0
                          if keep_checkpoint_every_n_hours else None)
    self._last_kept_time = 0.0
    self._protected: List[str] = []
    os.makedirs(model_dir, exist_ok=True)

  def save(self, step: int, network: torch.nn.Module,
           optimizer=None, ema=None,
           extra: Optional[Dict] = None) -> str:
    """Writes model.ckpt-<step>.pt; with EMA, stores AVERAGED weights."""
    if ema is not None:
      ema.swap_in()
    try:
      payload = {
          "global_step": step,
          "model_state": {k: v.detach().cpu().clone()
                          for k, v in network.state_dict().items()},
      }
    finally:
      if ema is not None:
        ema.swap_out()
    if optimizer is not None:
      payload["optimizer_state"] = optimizer.state_dict()
    if ema is not None:
      payload["ema_state"] = {
          "decay": ema.decay,
          "shadow": {k: v.detach().cpu().clone()
                     for k, v in ema.shadow.items()},
      }
      # The live (non-averaged) weights ride along so RESUME is exact even
      # though the canonical model_state holds averaged params.
      payload["raw_model_state"] = {
          k: v.detach().cpu().clone()
          for k, v in network.state_dict().items()}
    if extra:
      payload["extra"] = extra
    path = checkpoint_path(self._model_dir, step)
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)
    self._write_index(path)
    self._gc()
    return path

  def _write_index(self, path: str):
    index = os.path.join(self._model_dir, CHECKPOINT_INDEX)
    tmp = index + ".tmp"
    with open(tmp, "w") as f:
      f.write(os.path.basename(path) + "\n")
    os.replace(tmp, index)

  def _gc(self):
    ckpts = list_checkpoints(self._model_dir)
    now = time.time()
    while len(ckpts) > self._max_to_keep:
      victim = ckpts.pop(0)
      if victim in self._protected:
        continue
      if self._keep_every_s is not None and \
          now - self._last_kept_time >= self._keep_every_s:
        self._protected.append(victim)
        self._last_kept_time = now
        continue
      try:
        os.remove(victim)
      except OSError:
        pass

  @staticmethod
  def restore(path: str, network: torch.nn.Module, optimizer=None,
              ema=None, strict: bool = True) -> int:
    """Loads a checkpoint; returns the stored global_step."""
    payload = torch.load(path, map_location="cpu", weights_only=False)
    state = payload.get("raw_model_state", payload["model_state"]) \
        if ema is not None and "raw_model_state" in payload \
        else payload["model_state"]
    network.load_state_dict(state, strict=strict)
    if optimizer is not None and "optimizer_state" in payload:
      optimizer.load_state_dict(payload["optimizer_state"])
    if ema is not None and "ema_state" in payload:
      ema.load_state_dict(payload["ema_state"])
    return int(payload.get("global_step", 0))

  @staticmethod
  def load_model_state(path: str) -> Dict[str, torch.Tensor]:
    payload = torch.load(path, map_location="cpu", weights_only=False)
    return payload["model_state"]


def create_backup_checkpoint_for_eval(model_dir: str,
                                      backup_subdir: str =
                                      "current_eval_checkpoint",
                                      max_retries: int = 10,
                                      retry_sleep: float = 0.5
                                      ) -> Optional[str]:
  """Copies the newest checkpoint aside before evaluating.

  Guards against the trainer GC'ing the checkpoint mid-eval (reference
  `utils/train_eval.py:616-684`): retries while tmp files are present.
  """
  for _ in range(max_retries):
    latest = latest_checkpoint(model_dir)
    if latest is None:
      time.sleep(retry_sleep)
      continue
    if any(name.endswith(".tmp")
           for name in os.listdir(model_dir)):
      time.sleep(retry_sleep)
      continue
    backup_dir = os.path.join(model_dir, backup_subdir)
    os.makedirs(backup_dir, exist_ok=True)
    dst = os.path.join(backup_dir, os.path.basename(latest))
    if os.path.exists(dst):
      return dst
    try:
      tmp = dst + ".copytmp"
      shutil.copyfile(latest, tmp)
      os.replace(tmp, dst)
      # GC older backups.
      for name in os.listdir(backup_dir):
        p = os.path.join(backup_dir, name)
        if p != dst and _CKPT_RE.match(name):
          os.remove(p)
      return dst
    except OSError:
      time.sleep(retry_sleep)
  return None
