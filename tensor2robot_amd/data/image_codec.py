"""CPU image codecs: PNG (pure python/zlib) and JPEG (native extension).

The data pipeline decodes serialized image features declared with
`data_format` JPEG/PNG (reference `utils/tfdata.py:426-484`).  PNG is
implemented here over zlib; baseline JPEG encode/decode lives in the C++
extension (`tensor2robot_amd/cpp`), with the GPU decode path in HIP.
"""

from __future__ import annotations

import struct
import zlib
from typing import Optional

import numpy as np

_PNG_SIG = b"\x89PNG\r\n\x1a\n"


def _chunk(tag: bytes, payload: bytes) -> bytes:
  return (struct.pack(">I", len(payload)) + tag + payload +
          struct.pack(">I", zlib.crc32(tag + payload) & 0xFFFFFFFF))


def encode_png(image: np.ndarray) -> bytes:
  """Encodes HWC uint8/uint16 (C in {1,3,4}) to PNG bytes."""
  if image.ndim == 2:
    image = image[:, :, None]
  h, w, c = image.shape
  color_type = {1: 0, 2: 4, 3: 2, 4: 6}[c]
  if image.dtype == np.uint8:
    depth = 8
    raw = image
  elif image.dtype == np.uint16:
    depth = 16
    raw = image.astype(">u2")
  else:
    raise ValueError(f"PNG encode supports uint8/uint16, got {image.dtype}")
  ihdr = struct.pack(">IIBBBBB", w, h, depth, color_type, 0, 0, 0)
  # Filter type 0 (None) per scanline.
  rows = raw.reshape(h, -1).view(np.uint8).reshape(h, -1)
  scanlines = b"".join(b"\x00" + rows[i].tobytes() for i in range(h))
  idat = zlib.compress(scanlines, 6)
  return (_PNG_SIG + _chunk(b"IHDR", ihdr) + _chunk(b"IDAT", idat) +
          _chunk(b"IEND", b""))


def _unfilter(data: np.ndarray, h: int, stride: int, bpp: int) -> np.ndarray:
  """Reverses PNG scanline filters; returns (h, stride) uint8.

  The sequential Sub/Average/Paeth filters run through the native
  unfilter (data/native/example_codec.cpp, GIL-released) when the
  extension is built; this python loop is the reference semantics and
  the fallback."""
  native = _load_jpeg_native()
  if native is not None and hasattr(native, "png_unfilter"):
    return native.png_unfilter(data.tobytes(), h, stride, bpp)
  out = np.zeros((h, stride), dtype=np.uint8)
  pos = 0
  for y in range(h):
    ftype = data[pos]
    pos += 1
    row = data[pos: pos + stride].astype(np.int32)
    pos += stride
    prev = out[y - 1].astype(np.int32) if y > 0 else np.zeros(
        stride, np.int32)
    if ftype == 0:
      out[y] = row
    elif ftype == 2:  # Up
      out[y] = (row + prev) & 0xFF
    elif ftype in (1, 3, 4):
      cur = np.zeros(stride, np.int32)
      for x in range(stride):
        a = cur[x - bpp] if x >= bpp else 0
        b = prev[x]
        cc = prev[x - bpp] if x >= bpp else 0
        if ftype == 1:  # Sub
          pred = a
        elif ftype == 3:  # Average
          pred = (a + b) >> 1
        else:  # Paeth
          p = a + b - cc
          pa, pb, pc = abs(p - a), abs(p - b), abs(p - cc)
          pred = a if (pa <= pb and pa <= pc) else (b if pb <= pc else cc)
        cur[x] = (row[x] + pred) & 0xFF
      out[y] = cur
    else:
      raise ValueError(f"Unsupported PNG filter {ftype}")
  return out


def decode_png(data: bytes) -> np.ndarray:
  """Decodes PNG bytes to HWC uint8/uint16 ndarray."""
  if data[:8] != _PNG_SIG:
    raise ValueError("Not a PNG")
  pos = 8
  width = height = depth = color_type = None
  idat = bytearray()
  palette = None
  while pos < len(data):
    (length,) = struct.unpack(">I", data[pos: pos + 4])
    tag = data[pos + 4: pos + 8]
    payload = data[pos + 8: pos + 8 + length]
    pos += 12 + length
    if tag == b"IHDR":
      width, height, depth, color_type, comp, filt, interlace = \
          struct.unpack(">IIBBBBB", payload)
      if interlace:
        raise ValueError("Interlaced PNG unsupported")
    elif tag == b"IDAT":
      idat.extend(payload)
    elif tag == b"PLTE":
      palette = np.frombuffer(payload, np.uint8).reshape(-1, 3)
    elif tag == b"IEND":
      break
  channels = {0: 1, 2: 3, 3: 1, 4: 2, 6: 4}[color_type]
  raw = np.frombuffer(zlib.decompress(bytes(idat)), dtype=np.uint8)
  bits_pp = channels * depth
  bpp = max(1, bits_pp // 8)
  stride = (width * bits_pp + 7) // 8
  rows = _unfilter(raw, height, stride, bpp)
  if depth == 8:
    img = rows.reshape(height, width, channels)
  elif depth == 16:
    img = rows.reshape(height, -1).view(">u2").astype(np.uint16)
    img = img.reshape(height, width, channels)
  else:
    raise ValueError(f"PNG bit depth {depth} unsupported")
  if color_type == 3:  # palette
    img = palette[img[:, :, 0]]
  return img


_jpeg_native = None
_jpeg_import_error = None


def _load_jpeg_native():
  global _jpeg_native, _jpeg_import_error
  if _jpeg_native is None and _jpeg_import_error is None:
    try:
      from tensor2robot_amd.ops import _t2r_native  # built extension
      _jpeg_native = _t2r_native
    except ImportError as e:  # pragma: no cover
      _jpeg_import_error = e
  return _jpeg_native


def encode_jpeg(image: np.ndarray, quality: int = 90,
                restart_interval: int = 0) -> bytes:
  """restart_interval > 0 emits RSTn markers every N MCUs so the
  Huffman scan parallelizes at decode time (gpu_jpeg segment decode)."""
  native = _load_jpeg_native()
  if native is None:
    raise RuntimeError(
        f"JPEG codec extension not built: {_jpeg_import_error}. "
        "Run `python setup.py build_ext --inplace`.")
  image = np.ascontiguousarray(image, dtype=np.uint8)
  if image.ndim == 2:
    image = image[:, :, None]
  return native.encode_jpeg(image, quality, restart_interval)


def decode_jpeg(data: bytes) -> np.ndarray:
  native = _load_jpeg_native()
  if native is None:
    raise RuntimeError(
        f"JPEG codec extension not built: {_jpeg_import_error}. "
        "Run `python setup.py build_ext --inplace`.")
  return native.decode_jpeg(data)


def native_module():
  """The built _t2r_native extension (raises if missing)."""
  native = _load_jpeg_native()
  if native is None:
    raise RuntimeError(
        f"JPEG codec extension not built: {_jpeg_import_error}. "
        "Run `python setup.py build_ext --inplace`.")
  return native


def decode_image(data: bytes, data_format: Optional[str] = None) -> np.ndarray:
  """Decodes JPEG or PNG bytes (sniffs when data_format is None)."""
  if not data:
    raise ValueError("Empty image bytes")
  if data[:8] == _PNG_SIG:
    return decode_png(data)
  if data[:2] == b"\xff\xd8":
    return decode_jpeg(data)
  raise ValueError(
      f"Unrecognized image format (declared {data_format!r}, "
      f"magic {data[:4]!r})")


def encode_image(image: np.ndarray, data_format: str,
                 quality: int = 90) -> bytes:
  if data_format.upper() == "PNG":
    return encode_png(image)
  if data_format.upper() == "JPEG":
    return encode_jpeg(image, quality)
  raise ValueError(f"Unsupported data_format {data_format}")
