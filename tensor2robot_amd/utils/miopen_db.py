"""Packaged MIOpen perf-DB activation.

MIOpen's runtime algorithm search ("find") is a per-process lottery on
this pool: identical binaries measured 5.1k-7.7k img/s on the flagship
bench purely from which igemm kernels find drew (profiles/).  A tuned
user perf DB (produced once with MIOPEN_FIND_ENFORCE=SEARCH and
committed under tensor2robot_amd/miopen_db/) pins every MIOpen conv to
its searched-best kernel on any box of this image.

Call use_packaged_db() BEFORE the first convolution runs (MIOpen reads
MIOPEN_USER_DB_PATH at init).  The DB dir must be writable, so the
packaged files are copied to a temp dir first.
"""

from __future__ import annotations

import logging
import os
import shutil
import tempfile

_log = logging.getLogger(__name__)

DB_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "miopen_db")


def use_packaged_db() -> bool:
  """Points MIOPEN_USER_DB_PATH at a writable copy of the packaged DB."""
  if "MIOPEN_USER_DB_PATH" in os.environ:
    return True  # caller already configured MIOpen
  if not (os.path.isdir(DB_DIR) and os.listdir(DB_DIR)):
    return False
  target = tempfile.mkdtemp(prefix="t2r_miopen_db_")
  for name in os.listdir(DB_DIR):
    shutil.copy(os.path.join(DB_DIR, name), target)
  os.environ["MIOPEN_USER_DB_PATH"] = target
  _log.info("MIOpen user DB: %s (from packaged %s)", target, DB_DIR)
  return True
