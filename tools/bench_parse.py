"""CPU record->tensor parse throughput (native vs python wire decode).

  python tools/bench_parse.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import numpy as np
import torch

from tensor2robot_amd.data import example as ec
from tensor2robot_amd.data import parser as parser_mod
from tensor2robot_amd.specs import tensorspec_utils as tsu


def main():
  rng = np.random.RandomState(0)
  from tensor2robot_amd.data import image_codec
  img = (rng.rand(64, 64, 3) * 255).astype(np.uint8)
  png = image_codec.encode_png(img)
  records = [ec.encode_example({
      "img": [png],
      "action": rng.randn(10).astype(np.float32),
      "label": np.array([1.0], np.float32),
  }) for _ in range(64)]

  spec = tsu.TensorSpecStruct()
  spec["image"] = tsu.ExtendedTensorSpec((64, 64, 3), torch.uint8,
                                         name="img", data_format="PNG")
  spec["action"] = tsu.ExtendedTensorSpec((10,), torch.float32,
                                          name="action")
  labels = tsu.TensorSpecStruct()
  labels["label"] = tsu.ExtendedTensorSpec((1,), torch.float32,
                                           name="label")
  parse = parser_mod.create_parse_example_fn(spec, labels)
  for _ in range(3):
    parse(records)
  t0 = time.perf_counter()
  iters = 30
  for _ in range(iters):
    parse(records)
  dt = (time.perf_counter() - t0) / iters
  print(f"parse (PNG 64^2 + floats): {dt*1e3:.2f} ms/batch64 "
        f"({64/dt:.0f} rec/s)")

  spec2 = tsu.TensorSpecStruct()
  spec2["action"] = spec["action"]
  parse2 = parser_mod.create_parse_example_fn(spec2, labels)
  small = [ec.encode_example({
      "action": rng.randn(10).astype(np.float32),
      "label": np.array([1.0], np.float32)}) for _ in range(512)]
  for _ in range(3):
    parse2(small)
  t0 = time.perf_counter()
  for _ in range(50):
    parse2(small)
  dt = (time.perf_counter() - t0) / 50
  print(f"parse (floats only): {dt*1e3:.2f} ms/batch512 "
        f"({512/dt:.0f} rec/s)")


if __name__ == "__main__":
  main()
