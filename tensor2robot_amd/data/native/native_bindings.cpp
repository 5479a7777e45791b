// pybind11 bindings for the CPU-native data codecs (_t2r_native).

#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>

#include <string>
#include <vector>

namespace py = pybind11;

#include "jpeg_codec.h"

py::list parse_example_batch(py::sequence records);
py::list parse_sequence_example_batch(py::sequence records);
py::list read_tfrecord_file(const std::string& path, bool verify_crc);
py::array_t<uint8_t> png_unfilter(py::bytes data, int h, int stride,
                                  int bpp);

static py::bytes encode_jpeg(
    py::array_t<uint8_t, py::array::c_style | py::array::forcecast> image,
    int quality, int restart_interval) {
  const int ndim = image.ndim();
  if (ndim != 2 && ndim != 3)
    throw std::runtime_error("encode_jpeg: HxW or HxWxC uint8 expected");
  const int h = image.shape(0), w = image.shape(1);
  const int c = ndim == 2 ? 1 : image.shape(2);
  const uint8_t* data = image.data();
  std::vector<uint8_t> out;
  {
    py::gil_scoped_release release;
    out = t2r_jpeg::encode(data, h, w, c, quality, restart_interval);
  }
  return py::bytes((const char*)out.data(), out.size());
}

static py::array decode_jpeg(py::bytes data) {
  std::string buf = data;
  int h = 0, w = 0, c = 0;
  std::vector<uint8_t> img;
  {
    py::gil_scoped_release release;
    img = t2r_jpeg::decode((const uint8_t*)buf.data(), buf.size(), h, w,
                           c);
  }
  if (c == 1) {
    py::array_t<uint8_t> out({h, w});
    std::memcpy(out.mutable_data(), img.data(), img.size());
    return out;
  }
  py::array_t<uint8_t> out({h, w, c});
  std::memcpy(out.mutable_data(), img.data(), img.size());
  return out;
}

static py::dict decode_jpeg_coeffs(py::bytes data, int num_threads) {
  // Huffman/entropy decode only (the GPU decode handoff); the scan is
  // released from the GIL so batches parallelize across host threads,
  // and restart-marker streams additionally parallelize WITHIN one
  // image (num_threads segments decode concurrently).
  std::string buf = data;
  t2r_jpeg::CoeffImage ci;
  {
    py::gil_scoped_release release;
    ci = t2r_jpeg::decode_coeffs((const uint8_t*)buf.data(), buf.size(),
                                 num_threads);
  }
  py::dict out;
  out["height"] = ci.height;
  out["width"] = ci.width;
  out["ncomp"] = ci.ncomp;
  out["hmax"] = ci.hmax;
  out["vmax"] = ci.vmax;
  py::list comps;
  for (int c = 0; c < ci.ncomp; ++c) {
    auto& cp = ci.comps[c];
    py::dict d;
    d["hs"] = cp.hs;
    d["vs"] = cp.vs;
    py::array_t<int16_t> coeffs({cp.bh, cp.bw, 64});
    std::memcpy(coeffs.mutable_data(), cp.coeffs.data(),
                cp.coeffs.size() * sizeof(int16_t));
    d["coeffs"] = coeffs;
    py::array_t<uint16_t> q({64});
    std::memcpy(q.mutable_data(), ci.qt[cp.tq], 64 * sizeof(uint16_t));
    d["quant"] = q;
    comps.append(d);
  }
  out["comps"] = comps;
  return out;
}

PYBIND11_MODULE(_t2r_native, m) {
  m.doc() = "CPU-native codecs: baseline JPEG encode/decode";
  m.def("encode_jpeg", &encode_jpeg, py::arg("image"),
        py::arg("quality") = 90, py::arg("restart_interval") = 0);
  m.def("decode_jpeg", &decode_jpeg, py::arg("data"));
  m.def("decode_jpeg_coeffs", &decode_jpeg_coeffs, py::arg("data"),
        py::arg("num_threads") = 1);
  m.def("parse_example_batch", &parse_example_batch,
        py::arg("records"),
        "batch tf.Example wire decode (GIL-released scan)");
  m.def("parse_sequence_example_batch", &parse_sequence_example_batch,
        py::arg("records"),
        "batch tf.SequenceExample wire decode (GIL-released scan)");
  m.def("read_tfrecord_file", &read_tfrecord_file, py::arg("path"),
        py::arg("verify_crc") = true,
        "read a whole TFRecord shard (hardware CRC32C verify)");
  m.def("png_unfilter", &png_unfilter, py::arg("data"), py::arg("h"),
        py::arg("stride"), py::arg("bpp"),
        "PNG scanline unfilter (GIL-released)");
}
