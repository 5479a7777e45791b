// Shared declarations for the CPU-native baseline JPEG codec.
#pragma once

#include <cstddef>
#include <cstdint>
#include <vector>

namespace t2r_jpeg {

// restart_interval > 0 emits a DRI segment and an RSTn marker every
// that many MCUs — the entropy stream becomes independently decodable
// segments, which is what lets decode_coeffs() parallelize the
// bit-serial Huffman scan inside ONE image.
std::vector<uint8_t> encode(const uint8_t* rgb, int h, int w, int channels,
                            int quality, int restart_interval = 0);
std::vector<uint8_t> decode(const uint8_t* data, size_t size, int& out_h,
                            int& out_w, int& out_c);

// Quantized DCT coefficients + metadata (the GPU decode handoff:
// Huffman on host threads, dequant/IDCT/upsample/color in HIP).
struct CoeffImage {
  int height = 0, width = 0, ncomp = 0, hmax = 1, vmax = 1;
  uint16_t qt[4][64] = {};
  struct Comp {
    int hs = 1, vs = 1, tq = 0;
    int bw = 0, bh = 0;              // blocks across / down
    std::vector<int16_t> coeffs;     // [bh][bw][64] natural order
  } comps[4];
};
// num_threads > 1 decodes restart-marker segments concurrently (falls
// back to the sequential scan when the stream has no restart markers).
CoeffImage decode_coeffs(const uint8_t* data, size_t size,
                         int num_threads = 1);

}  // namespace t2r_jpeg
