// Fused spatial softmax (soft arg-max) for NHWC bf16 feature maps.
//
// Reference layers/spatial_softmax.py:29-89 composes reshape + softmax
// + a [HW,2] matmul (5-6 kernels with dtype casts).  SURVEY 2.10 item
// 6 calls this "a natural single fused kernel": forward runs ONE
// online-softmax pass per (image, channel) accumulating max / sum /
// x-expectation / y-expectation in registers, then one short pass to
// emit the softmax map; backward is one fused pass.
//
// Layout: channels_last [N,C,H,W] -> per pixel the C channels are
// contiguous, so a block's lanes (one channel each) read coalesced
// 2B*C rows.  f32 accumulation, bf16 tensors.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 ssbf16_t;

__device__ __forceinline__ float ss_b2f(ssbf16_t v) {
  return __bfloat162float(v);
}

// grid.x = N, block = THREADS (>= C, multiple of 64); lane c handles
// channel c of image n.
extern "C" __global__ void __launch_bounds__(256)
spatial_softmax_fwd_kernel(const ssbf16_t* __restrict__ x,
                           ssbf16_t* __restrict__ points,  // [N, 2C]
                           ssbf16_t* __restrict__ map,     // [N,H,W,C]
                           int N, int C, int H, int W, float inv_t) {
  const int c = threadIdx.x;
  const int n = blockIdx.x;
  if (c >= C) return;
  const long base = (long)n * H * W * C + c;
  const float gx_step = (W > 1) ? 2.0f / (W - 1) : 0.0f;
  const float gy_step = (H > 1) ? 2.0f / (H - 1) : 0.0f;

  // Online softmax with coordinate expectations.
  float m = -3.4e38f, s = 0.0f, sx = 0.0f, sy = 0.0f;
  long idx = base;
  for (int h = 0; h < H; ++h) {
    const float gy = (H > 1) ? -1.0f + gy_step * h : 0.0f;
    for (int w = 0; w < W; ++w, idx += C) {
      const float gx = (W > 1) ? -1.0f + gx_step * w : 0.0f;
      const float v = ss_b2f(x[idx]) * inv_t;
      if (v > m) {
        const float scale = __expf(m - v);
        s *= scale; sx *= scale; sy *= scale;
        m = v;
      }
      const float e = __expf(v - m);
      s += e;
      sx += e * gx;
      sy += e * gy;
    }
  }
  const float inv_s = 1.0f / s;
  // Reference layout: [N, 2C] = channel-major (x, y) pairs
  // ([x0 y0 x1 y1 ...]).
  points[(long)n * 2 * C + 2 * c] = __float2bfloat16(sx * inv_s);
  points[(long)n * 2 * C + 2 * c + 1] = __float2bfloat16(sy * inv_s);

  idx = base;
  for (int p = 0; p < H * W; ++p, idx += C) {
    map[idx] = __float2bfloat16(__expf(ss_b2f(x[idx]) * inv_t - m)
                                * inv_s);
  }
}

// dL/dx[p,c] = sm[p,c] * ((gx[p]-px)*dpx + (gy[p]-py)*dpy
//                         + dmap[p,c] - sum_q sm[q,c]*dmap[q,c]) / T
extern "C" __global__ void __launch_bounds__(256)
spatial_softmax_bwd_kernel(const ssbf16_t* __restrict__ map,
                           const ssbf16_t* __restrict__ points,
                           const ssbf16_t* __restrict__ dpoints,  // or null
                           const ssbf16_t* __restrict__ dmap,     // or null
                           ssbf16_t* __restrict__ dx,
                           int N, int C, int H, int W, float inv_t) {
  const int c = threadIdx.x;
  const int n = blockIdx.x;
  if (c >= C) return;
  const long base = (long)n * H * W * C + c;
  const float gx_step = (W > 1) ? 2.0f / (W - 1) : 0.0f;
  const float gy_step = (H > 1) ? 2.0f / (H - 1) : 0.0f;

  float dpx = 0.0f, dpy = 0.0f;
  if (dpoints != nullptr) {
    dpx = ss_b2f(dpoints[(long)n * 2 * C + 2 * c]);
    dpy = ss_b2f(dpoints[(long)n * 2 * C + 2 * c + 1]);
  }
  const float px = ss_b2f(points[(long)n * 2 * C + 2 * c]);
  const float py = ss_b2f(points[(long)n * 2 * C + 2 * c + 1]);

  float dot = 0.0f;
  if (dmap != nullptr) {
    long idx = base;
    for (int p = 0; p < H * W; ++p, idx += C) {
      dot += ss_b2f(map[idx]) * ss_b2f(dmap[idx]);
    }
  }

  long idx = base;
  for (int h = 0; h < H; ++h) {
    const float gy = (H > 1) ? -1.0f + gy_step * h : 0.0f;
    for (int w = 0; w < W; ++w, idx += C) {
      const float gx = (W > 1) ? -1.0f + gx_step * w : 0.0f;
      float u = (gx - px) * dpx + (gy - py) * dpy;
      if (dmap != nullptr) u += ss_b2f(dmap[idx]) - dot;
      dx[idx] = __float2bfloat16(ss_b2f(map[idx]) * u * inv_t);
    }
  }
}

static void ss_check(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16,
              name, ": bf16 CUDA required");
}

std::vector<at::Tensor> spatial_softmax_fwd(at::Tensor x, double temp) {
  ss_check(x, "spatial_softmax_fwd");
  TORCH_CHECK(x.dim() == 4 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "spatial_softmax_fwd: [N,C,H,W] channels_last required");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C <= 256, "spatial_softmax_fwd: C <= 256");
  auto points = at::empty({N, 2 * C}, x.options());
  auto map = at::empty({N, C, H, W},
                       x.options().memory_format(
                           at::MemoryFormat::ChannelsLast));
  const int threads = std::max(64, ((C + 63) / 64) * 64);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(spatial_softmax_fwd_kernel, dim3(N), dim3(threads),
                     0, stream.stream(),
                     (const ssbf16_t*)x.data_ptr(),
                     (ssbf16_t*)points.data_ptr(),
                     (ssbf16_t*)map.data_ptr(),
                     N, C, H, W, (float)(1.0 / temp));
  return {points, map};
}

at::Tensor spatial_softmax_bwd(at::Tensor map, at::Tensor points,
                               c10::optional<at::Tensor> dpoints,
                               c10::optional<at::Tensor> dmap,
                               double temp) {
  ss_check(map, "spatial_softmax_bwd");
  const int N = map.size(0), C = map.size(1), H = map.size(2),
            W = map.size(3);
  auto dx = at::empty_like(map);
  const int threads = std::max(64, ((C + 63) / 64) * 64);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      spatial_softmax_bwd_kernel, dim3(N), dim3(threads), 0,
      stream.stream(), (const ssbf16_t*)map.data_ptr(),
      (const ssbf16_t*)points.data_ptr(),
      dpoints.has_value() ? (const ssbf16_t*)dpoints->data_ptr()
                          : nullptr,
      dmap.has_value() ? (const ssbf16_t*)dmap->data_ptr() : nullptr,
      (ssbf16_t*)dx.data_ptr(), N, C, H, W, (float)(1.0 / temp));
  return dx;
}
