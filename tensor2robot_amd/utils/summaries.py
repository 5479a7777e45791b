"""Scalar summary writer (tensorboard-style observability).

The reference's observability is TF summaries (`SURVEY.md §5.1`).  Here:
scalars append to `events.jsonl` inside the log dir — one JSON object per
record {step, tag, value, wall_time} — easily consumed by notebooks or
converted to TensorBoard.  The abstraction point matches the reference's
`add_summaries` hook so composed models (MAML) can emit after the fact.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional


class SummaryWriter:

  def __init__(self, log_dir: str, filename: str = "events.jsonl",
               tensorboard: bool = False):
    os.makedirs(log_dir, exist_ok=True)
    self._path = os.path.join(log_dir, filename)
    self._file = open(self._path, "a")
    self._tb = None
    if tensorboard or os.environ.get("T2R_TENSORBOARD_EVENTS"):
      from tensor2robot_amd.utils import tb_events
      self._tb = tb_events.TBEventWriter(log_dir)

  def add_scalar(self, tag: str, value: float, step: int):
    rec = {"step": int(step), "tag": tag, "value": float(value),
           "wall_time": time.time()}
    self._file.write(json.dumps(rec) + "\n")
    if self._tb is not None:
      self._tb.add_scalar(tag, value, step)

  def add_scalars(self, scalars: Dict[str, float], step: int):
    for tag, value in scalars.items():
      self.add_scalar(tag, value, step)

  def flush(self):
    self._file.flush()

  def close(self):
    if self._file:
      self._file.flush()
      self._file.close()
      self._file = None
    if self._tb is not None:
      self._tb.close()
      self._tb = None

  def __enter__(self):
    return self

  def __exit__(self, *exc):
    self.close()


def read_events(log_dir: str, filename: str = "events.jsonl"):
  path = os.path.join(log_dir, filename)
  if not os.path.exists(path):
    return []
  out = []
  with open(path) as f:
    for line in f:
      line = line.strip()
      if line:
        out.append(json.loads(line))
  return out
