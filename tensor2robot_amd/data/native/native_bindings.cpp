// pybind11 bindings for the CPU-native data codecs (_t2r_native).

#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>

#include <vector>

namespace py = pybind11;

namespace t2r_jpeg {
std::vector<uint8_t> encode(const uint8_t* rgb, int h, int w, int channels,
                            int quality);
std::vector<uint8_t> decode(const uint8_t* data, size_t size, int& out_h,
                            int& out_w, int& out_c);
}  // namespace t2r_jpeg

static py::bytes encode_jpeg(
    py::array_t<uint8_t, py::array::c_style | py::array::forcecast> image,
    int quality) {
  if (image.ndim() == 2) {
    auto out = t2r_jpeg::encode(image.data(), image.shape(0),
                                image.shape(1), 1, quality);
    return py::bytes((const char*)out.data(), out.size());
  }
  if (image.ndim() != 3)
    throw std::runtime_error("encode_jpeg: HxW or HxWxC uint8 expected");
  auto out = t2r_jpeg::encode(image.data(), image.shape(0),
                              image.shape(1), image.shape(2), quality);
  return py::bytes((const char*)out.data(), out.size());
}

static py::array decode_jpeg(py::bytes data) {
  std::string buf = data;
  int h = 0, w = 0, c = 0;
  auto img = t2r_jpeg::decode((const uint8_t*)buf.data(), buf.size(), h, w,
                              c);
  if (c == 1) {
    py::array_t<uint8_t> out({h, w});
    std::memcpy(out.mutable_data(), img.data(), img.size());
    return out;
  }
  py::array_t<uint8_t> out({h, w, c});
  std::memcpy(out.mutable_data(), img.data(), img.size());
  return out;
}

PYBIND11_MODULE(_t2r_native, m) {
  m.doc() = "CPU-native codecs: baseline JPEG encode/decode";
  m.def("encode_jpeg", &encode_jpeg, py::arg("image"),
        py::arg("quality") = 90);
  m.def("decode_jpeg", &decode_jpeg, py::arg("data"));
}
