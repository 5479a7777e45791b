// GPU JPEG decode stages (gfx950): dequant + 8x8 IDCT, then
// upsample + YCbCr->RGB, as two batched kernels.
//
// The bit-serial Huffman scan runs on host threads
// (data/native/jpeg_codec.cpp decode_coeffs, GIL-released); every
// numeric stage runs here.  Replaces the reference's CPU
// tf.image.decode_image in the input pipeline
// (`utils/tfdata.py:426-484`) — SURVEY §2.10 item 7.
//
// Batch layout: all images of a batch share geometry (training data
// is fixed-shape), so a component's coefficients stack to
// [N * bh * bw, 64] int16 and decode in one dispatch.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

// One thread = one 8x8 block: 64 int16 loads (4x uint4), dequant by
// the (uniform, LDS-cached) quant table, separable IDCT in registers,
// 8 x 32-B row stores into the component plane (+128 level shift).
extern "C" __global__ void __launch_bounds__(256)
jpeg_idct_kernel(const int16_t* __restrict__ coeffs,
                 const int* __restrict__ quant,
                 float* __restrict__ plane,
                 long nblocks, int bw, int bh) {
  __shared__ float s_q[64];
  __shared__ float s_cos[64];
  if (threadIdx.x < 64) {
    s_q[threadIdx.x] = (float)quant[threadIdx.x];
    const int u = threadIdx.x >> 3, x = threadIdx.x & 7;
    s_cos[threadIdx.x] =
        cosf((2 * x + 1) * u * (float)M_PI / 16.0f) *
        ((u == 0) ? 0.70710678f : 1.0f);
  }
  __syncthreads();
  const int pw = bw * 8;
  const long ppi = (long)pw * bh * 8;      // plane elems per image
  const long bpi = (long)bw * bh;          // blocks per image
  for (long b = (long)blockIdx.x * 256 + threadIdx.x; b < nblocks;
       b += (long)gridDim.x * 256) {
    const int16_t* src = coeffs + b * 64;
    float f[64];
#pragma unroll
    for (int i = 0; i < 64; ++i) f[i] = (float)src[i] * s_q[i];
    // cols: tmp[y][v] = sum_u cu * f[u][v] * cos[u][y]
    float tmp[64];
#pragma unroll
    for (int v = 0; v < 8; ++v) {
#pragma unroll
      for (int y = 0; y < 8; ++y) {
        float s = 0.f;
#pragma unroll
        for (int u = 0; u < 8; ++u) s += f[u * 8 + v] * s_cos[u * 8 + y];
        tmp[y * 8 + v] = s;
      }
    }
    const long img = b / bpi;
    const long brest = b % bpi;
    const int by = (int)(brest / bw), bx = (int)(brest % bw);
    float* out = plane + img * ppi + ((long)by * 8) * pw + bx * 8;
#pragma unroll
    for (int y = 0; y < 8; ++y) {
      float row[8];
#pragma unroll
      for (int x = 0; x < 8; ++x) {
        float s = 0.f;
#pragma unroll
        for (int u = 0; u < 8; ++u)
          s += tmp[y * 8 + u] * s_cos[u * 8 + x];
        row[x] = 0.25f * s + 128.0f;
      }
      *reinterpret_cast<float4*>(out + (long)y * pw) =
          make_float4(row[0], row[1], row[2], row[3]);
      *reinterpret_cast<float4*>(out + (long)y * pw + 4) =
          make_float4(row[4], row[5], row[6], row[7]);
    }
  }
}

__device__ __forceinline__ uint8_t jpeg_clamp8(float v) {
  return (uint8_t)min(max(__float2int_rn(v), 0), 255);
}

// One thread = 4 consecutive output pixels of one row: nearest-sample
// the chroma planes, YCbCr -> RGB, one 12-B store (3 dwords).
extern "C" __global__ void __launch_bounds__(256)
jpeg_color_kernel(const float* __restrict__ py,
                  const float* __restrict__ pcb,
                  const float* __restrict__ pcr,
                  uint8_t* __restrict__ out,
                  int N, int H, int W,
                  int pwy, int phy, int pwc, int phc,
                  int hs_y, int vs_y, int hs_c, int vs_c,
                  int hmax, int vmax) {
  const int wq = (W + 3) / 4;
  const long cells = (long)N * H * wq;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int xq = (int)(i % wq);
    const int y = (int)((i / wq) % H);
    const int n = (int)(i / ((long)H * wq));
    const float* yb = py + (long)n * pwy * phy;
    const float* cbb = pcb + (long)n * pwc * phc;
    const float* crb = pcr + (long)n * pwc * phc;
    uint8_t px[12];
    const int x0 = xq * 4;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      const int x = min(x0 + d, W - 1);
      const float Y =
          yb[(long)(y * vs_y / vmax) * pwy + (x * hs_y / hmax)];
      const float cb =
          cbb[(long)(y * vs_c / vmax) * pwc + (x * hs_c / hmax)] - 128.f;
      const float cr =
          crb[(long)(y * vs_c / vmax) * pwc + (x * hs_c / hmax)] - 128.f;
      px[d * 3 + 0] = jpeg_clamp8(Y + 1.402f * cr);
      px[d * 3 + 1] = jpeg_clamp8(Y - 0.344136f * cb - 0.714136f * cr);
      px[d * 3 + 2] = jpeg_clamp8(Y + 1.772f * cb);
    }
    uint8_t* dst = out + (((long)n * H + y) * W + x0) * 3;
    const int nbytes = (min(x0 + 4, W) - x0) * 3;
    if (nbytes == 12 && (((size_t)dst) & 3) == 0) {
      const uint32_t* s32 = reinterpret_cast<const uint32_t*>(px);
      uint32_t* d32 = reinterpret_cast<uint32_t*>(dst);
      d32[0] = s32[0];
      d32[1] = s32[1];
      d32[2] = s32[2];
    } else {
      for (int j = 0; j < nbytes; ++j) dst[j] = px[j];
    }
  }
}

// Grayscale: one thread = 4 pixels, clamp the Y plane.
extern "C" __global__ void __launch_bounds__(256)
jpeg_gray_kernel(const float* __restrict__ py,
                 uint8_t* __restrict__ out,
                 int N, int H, int W, int pwy, int phy) {
  const int wq = (W + 3) / 4;
  const long cells = (long)N * H * wq;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int xq = (int)(i % wq);
    const int y = (int)((i / wq) % H);
    const int n = (int)(i / ((long)H * wq));
    const float* yb = py + (long)n * pwy * phy;
    const int x0 = xq * 4;
    for (int d = 0; d < 4 && x0 + d < W; ++d) {
      out[((long)n * H + y) * W + x0 + d] =
          jpeg_clamp8(yb[(long)y * pwy + x0 + d]);
    }
  }
}

static int jg_grid(long cells) {
  long wgs = (cells + 255) / 256;
  if (wgs > 4096) wgs = 4096;
  if (wgs < 1) wgs = 1;
  return (int)wgs;
}

at::Tensor jpeg_idct(at::Tensor coeffs, at::Tensor quant, int64_t bh,
                     int64_t bw) {
  TORCH_CHECK(coeffs.is_cuda() && coeffs.scalar_type() == at::kShort,
              "jpeg_idct: int16 CUDA coeffs required");
  TORCH_CHECK(quant.is_cuda() &&
              quant.scalar_type() == at::kInt && quant.numel() == 64,
              "jpeg_idct: int32[64] CUDA quant required");
  coeffs = coeffs.contiguous();
  const long nblocks = coeffs.numel() / 64;
  TORCH_CHECK(nblocks % (bh * bw) == 0, "jpeg_idct: batch shape");
  const long n = nblocks / (bh * bw);
  auto plane = at::empty({n, bh * 8, bw * 8},
                         coeffs.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(jpeg_idct_kernel, dim3(jg_grid(nblocks)), dim3(256),
                     0, stream.stream(),
                     (const int16_t*)coeffs.data_ptr(),
                     (const int*)quant.data_ptr(),
                     plane.data_ptr<float>(), nblocks, (int)bw, (int)bh);
  return plane;
}

at::Tensor jpeg_color(at::Tensor py, at::Tensor pcb, at::Tensor pcr,
                      int64_t H, int64_t W,
                      int64_t hs_y, int64_t vs_y, int64_t hs_c,
                      int64_t vs_c, int64_t hmax, int64_t vmax) {
  TORCH_CHECK(py.is_cuda() && py.scalar_type() == at::kFloat,
              "jpeg_color: f32 planes required");
  const int n = py.size(0);
  auto out = at::empty({(long)n, H, W, 3},
                       py.options().dtype(at::kByte));
  const long cells = (long)n * H * ((W + 3) / 4);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(jpeg_color_kernel, dim3(jg_grid(cells)), dim3(256),
                     0, stream.stream(), py.data_ptr<float>(),
                     pcb.data_ptr<float>(), pcr.data_ptr<float>(),
                     (uint8_t*)out.data_ptr(),
                     n, (int)H, (int)W,
                     (int)py.size(2), (int)py.size(1),
                     (int)pcb.size(2), (int)pcb.size(1),
                     (int)hs_y, (int)vs_y, (int)hs_c, (int)vs_c,
                     (int)hmax, (int)vmax);
  return out;
}

at::Tensor jpeg_gray(at::Tensor py, int64_t H, int64_t W) {
  TORCH_CHECK(py.is_cuda() && py.scalar_type() == at::kFloat,
              "jpeg_gray: f32 plane required");
  const int n = py.size(0);
  auto out = at::empty({(long)n, H, W, 1},
                       py.options().dtype(at::kByte));
  const long cells = (long)n * H * ((W + 3) / 4);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(jpeg_gray_kernel, dim3(jg_grid(cells)), dim3(256),
                     0, stream.stream(), py.data_ptr<float>(),
                     (uint8_t*)out.data_ptr(), n, (int)H, (int)W,
                     (int)py.size(2), (int)py.size(1));
  return out;
}
