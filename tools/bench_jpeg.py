"""JPEG decode throughput: GPU path vs CPU codec, from real TFRecords.

Writes a TFRecord of 512x640 encoded JPEGs, then measures:
  * CPU: single-thread and thread-pool decode_jpeg
  * GPU: decode_jpeg_batch (host huffman threads + HIP idct/color)

  python tools/bench_jpeg.py [--images 64] [--batch 32]
"""

import argparse
import concurrent.futures
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from tensor2robot_amd.data import image_codec
from tensor2robot_amd.data import tfrecord


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--images", type=int, default=64)
  p.add_argument("--batch", type=int, default=32)
  p.add_argument("--height", type=int, default=512)
  p.add_argument("--width", type=int, default=640)
  args = p.parse_args()

  rng = np.random.RandomState(0)
  h, w = args.height, args.width
  yy, xx = np.mgrid[0:h, 0:w]
  path = os.path.join(tempfile.mkdtemp(), "jpegs.tfrecord")
  with tfrecord.TFRecordWriter(path) as wr:
    for i in range(args.images):
      base = (128 + 70 * np.sin(xx / (9.0 + i)) *
              np.cos(yy / (6.0 + i)))[..., None]
      img = np.clip(base + rng.randint(-25, 25, (h, w, 3)),
                    0, 255).astype(np.uint8)
      wr.write(image_codec.encode_jpeg(img, 90))
  records = list(tfrecord.read_records(path))
  print(f"{len(records)} records, avg {np.mean([len(r) for r in records])/1e3:.0f} KB")

  # CPU single thread
  t0 = time.perf_counter()
  for r in records[:16]:
    image_codec.decode_jpeg(r)
  cpu1 = 16 / (time.perf_counter() - t0)
  # CPU pool
  pool = concurrent.futures.ThreadPoolExecutor(16)
  t0 = time.perf_counter()
  list(pool.map(image_codec.decode_jpeg, records))
  cpuN = len(records) / (time.perf_counter() - t0)
  print(f"CPU decode: single {cpu1:7.1f} img/s | pool {cpuN:7.1f} img/s")

  if torch.cuda.is_available():
    from tensor2robot_amd.data import gpu_jpeg
    bs = args.batch
    batches = [records[i:i + bs] for i in range(0, len(records), bs)]
    gpu_jpeg.decode_jpeg_batch(batches[0], device="cuda")  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for b in batches:
      gpu_jpeg.decode_jpeg_batch(b, device="cuda")
    torch.cuda.synchronize()
    gpu = len(records) / (time.perf_counter() - t0)
    # Huffman-only rate (the host-side share of the GPU path).
    t0 = time.perf_counter()
    gpu_jpeg._huffman_batch(records)
    huff = len(records) / (time.perf_counter() - t0)
    print(f"GPU decode: {gpu:7.1f} img/s (host huffman alone "
          f"{huff:7.1f} img/s)")


if __name__ == "__main__":
  main()
