// Fused image preprocessing for the training input pipeline (gfx950).
//
// One stats pass + one apply pass replace the reference's chain of crop /
// convert_image_dtype / brightness / saturation / contrast ops
// (`preprocessors/image_transformations.py`, `distortion.py`) and their
// mean-reduction kernels (7% of baseline step time, see profiles/).
//
//   raw:  uint8 NHWC [N, H, W, 3]      (as parsed / decoded)
//   out:  bf16 or f32 NHWC [N, th, tw, 3], values in [0, 1]
//
// Math (matching image_transformations.ApplyPhotometricImageDistortions
// order: brightness -> saturation -> contrast -> clamp):
//   x  = v/255 + delta_b
//   g  = mean_c(x);  x = g + (x - g) * f_sat
//   m_c = per-image per-channel mean of x over the crop
//       = f_sat * mean(v_c)/255 + (1 - f_sat) * mean(gray_v)/255 + delta_b
//   x  = (x - m_c) * f_con + m_c;  clamp [0, 1]
// so a single uint8 integer-sum pass over the crop provides every mean.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 bf16_t;

// Per-image per-channel uint32 pixel sums over the crop window.
extern "C" __global__ void __launch_bounds__(256)
preprocess_stats_kernel(const unsigned char* __restrict__ raw,
                        unsigned int* __restrict__ sums,  // [N, 4]
                        int H, int W, int oy, int ox, int th, int tw) {
  const int n = blockIdx.y;
  const long npix = (long)th * tw;
  unsigned int local[4] = {0u, 0u, 0u, 0u};
  for (long p = (long)blockIdx.x * 256 + threadIdx.x; p < npix;
       p += (long)gridDim.x * 256) {
    const int y = (int)(p / tw), x = (int)(p % tw);
    const long off = (((long)n * H + oy + y) * W + ox + x) * 3;
    unsigned int r = raw[off], g = raw[off + 1], b = raw[off + 2];
    local[0] += r;
    local[1] += g;
    local[2] += b;
    local[3] += r + g + b;  // 3 * gray
  }
  __shared__ unsigned int lds[4][256];
#pragma unroll
  for (int c = 0; c < 4; ++c) lds[c][threadIdx.x] = local[c];
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
#pragma unroll
      for (int c = 0; c < 4; ++c)
        lds[c][threadIdx.x] += lds[c][threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x < 4) atomicAdd(&sums[n * 4 + threadIdx.x],
                                 lds[threadIdx.x][0]);
}

template <typename OutT>
__device__ __forceinline__ OutT to_out(float v);
template <> __device__ __forceinline__ float to_out<float>(float v) {
  return v;
}
template <> __device__ __forceinline__ bf16_t to_out<bf16_t>(float v) {
  return __float2bfloat16(v);
}

template <typename OutT>
__global__ void __launch_bounds__(256)
preprocess_apply_kernel(const unsigned char* __restrict__ raw,
                        OutT* __restrict__ out,
                        const unsigned int* __restrict__ sums,
                        const float* __restrict__ delta_b,   // [N]
                        const float* __restrict__ f_sat,     // [N]
                        const float* __restrict__ f_con,     // [N]
                        int H, int W, int oy, int ox, int th, int tw,
                        int distort) {
  const int n = blockIdx.y;
  const long npix = (long)th * tw;
  float db = 0.f, fs = 1.f, fc = 1.f, m[3] = {0.f, 0.f, 0.f};
  if (distort) {
    db = delta_b[n];
    fs = f_sat[n];
    fc = f_con[n];
    const float inv = 1.0f / (255.0f * (float)npix);
    const float gray_mean = (float)sums[n * 4 + 3] * inv * (1.0f / 3.0f);
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      const float mc = (float)sums[n * 4 + c] * inv;
      m[c] = fs * mc + (1.f - fs) * gray_mean + db;
    }
  }
  for (long p = (long)blockIdx.x * 256 + threadIdx.x; p < npix;
       p += (long)gridDim.x * 256) {
    const int y = (int)(p / tw), x = (int)(p % tw);
    const long src = (((long)n * H + oy + y) * W + ox + x) * 3;
    const long dst = ((long)n * npix + p) * 3;
    float v[3];
#pragma unroll
    for (int c = 0; c < 3; ++c) v[c] = (float)raw[src + c] * (1.f / 255.f);
    if (distort) {
      const float gray = (v[0] + v[1] + v[2]) * (1.f / 3.f) + db;
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        float t = gray + (v[c] + db - gray) * fs;     // brightness+saturation
        t = (t - m[c]) * fc + m[c];                   // contrast
        v[c] = fminf(fmaxf(t, 0.f), 1.f);
      }
    }
#pragma unroll
    for (int c = 0; c < 3; ++c) out[dst + c] = to_out<OutT>(v[c]);
  }
}

at::Tensor fused_preprocess(at::Tensor raw, int64_t oy, int64_t ox,
                            int64_t th, int64_t tw,
                            c10::optional<at::Tensor> delta_b,
                            c10::optional<at::Tensor> f_sat,
                            c10::optional<at::Tensor> f_con,
                            bool out_bf16) {
  TORCH_CHECK(raw.is_cuda() && raw.scalar_type() == at::kByte,
              "raw must be uint8 on GPU");
  TORCH_CHECK(raw.dim() == 4 && raw.size(3) == 3,
              "raw must be [N,H,W,3]");
  TORCH_CHECK(raw.is_contiguous(), "raw must be contiguous NHWC");
  const int N = (int)raw.size(0), H = (int)raw.size(1),
            W = (int)raw.size(2);
  TORCH_CHECK(oy >= 0 && ox >= 0 && oy + th <= H && ox + tw <= W,
              "crop window out of bounds");
  const bool distort = delta_b.has_value();
  auto stream = at::cuda::getCurrentCUDAStream();
  auto out = at::empty({N, th, tw, 3},
                       raw.options().dtype(out_bf16 ? at::kBFloat16
                                                    : at::kFloat));
  at::Tensor sums;
  const long npix = (long)th * tw;
  int grid_x = (int)std::min<long>((npix + 255) / 256, 512);
  if (distort) {
    sums = at::zeros({N, 4}, raw.options().dtype(at::kInt));
    hipLaunchKernelGGL(preprocess_stats_kernel, dim3(grid_x, N), dim3(256),
                       0, stream.stream(), raw.data_ptr<unsigned char>(),
                       reinterpret_cast<unsigned int*>(sums.data_ptr()),
                       H, W, (int)oy, (int)ox, (int)th, (int)tw);
  }
  const unsigned int* sums_ptr = distort
      ? reinterpret_cast<const unsigned int*>(sums.data_ptr()) : nullptr;
  const float* db = distort ? delta_b->data_ptr<float>() : nullptr;
  const float* fs = distort ? f_sat->data_ptr<float>() : nullptr;
  const float* fc = distort ? f_con->data_ptr<float>() : nullptr;
  if (out_bf16) {
    hipLaunchKernelGGL(preprocess_apply_kernel<bf16_t>, dim3(grid_x, N),
                       dim3(256), 0, stream.stream(),
                       raw.data_ptr<unsigned char>(),
                       reinterpret_cast<bf16_t*>(out.data_ptr()), sums_ptr,
                       db, fs, fc, H, W, (int)oy, (int)ox, (int)th,
                       (int)tw, distort ? 1 : 0);
  } else {
    hipLaunchKernelGGL(preprocess_apply_kernel<float>, dim3(grid_x, N),
                       dim3(256), 0, stream.stream(),
                       raw.data_ptr<unsigned char>(),
                       out.data_ptr<float>(), sums_ptr, db, fs, fc, H, W,
                       (int)oy, (int)ox, (int)th, (int)tw,
                       distort ? 1 : 0);
  }
  return out;
}
