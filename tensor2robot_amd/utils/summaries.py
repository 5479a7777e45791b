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

  def __init__(self, log_dir: str, filename: str = "events.jsonl"):
    os.makedirs(log_dir, exist_ok=True)
    self._path = os.path.join(log_dir, filename)
    self._file = open(self._path, "a")

  def add_scalar(self, tag: str, value: float, step: int):
    rec = {"step": int(step), "tag": tag, "value": float(value),
           "wall_time": time.time()}
    self._file.write(json.dumps(rec) + "\n")

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
