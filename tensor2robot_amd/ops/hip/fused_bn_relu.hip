// Fused training BatchNorm + ReLU for NHWC bf16 tensors (gfx950 / CDNA4).
//
// Replaces the MIOpen BN kernel chain (MeanVariance, FinalMeanVariance,
// Norm) + separate ReLU clamp + their backward counterparts with 2 forward
// and 2 backward kernels.  The tensor is viewed as a flat [M, C] matrix
// (NHWC channels-last: C contiguous, M = N*H*W), so every wave issues
// 16-byte (bf16x8) loads that are perfectly coalesced.
//
// Design notes (cdna_hip_programming.md):
//  * wave64; 256-thread workgroups; each thread owns 8 consecutive channels
//    (one uint4 load) => C must be a multiple of 8 (python falls back to
//    torch otherwise).
//  * grid is oversubscribed (>> 256 CUs) with a grid-stride loop; LDS
//    tree-reduce inside the workgroup, one float atomicAdd per channel per
//    workgroup to the global partial buffers (few thousand atomics total).
//  * stats/params/accumulators are fp32; data is bf16.
//  * ReLU is fused into the normalize pass; backward masks with y > 0 so no
//    separate mask tensor is stored.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdint>

#define T2R_CHECK(cond, msg) TORCH_CHECK(cond, msg)

typedef __hip_bfloat16 bf16_t;

__device__ __forceinline__ float bf2f(bf16_t v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ bf16_t f2bf(float v) {
  return __float2bfloat16(v);
}

union Vec8 {
  uint4 raw;
  bf16_t v[8];
};

// ---------------------------------------------------------------------------
// Forward stats: partial per-channel sum / sum-of-squares.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
bn_stats_kernel(const bf16_t* __restrict__ x, float* __restrict__ psum,
                float* __restrict__ psq, long M, int C) {
  const int tpr = C >> 3;              // threads per row
  const int rows = 256 / tpr;          // rows handled per wg iteration
  const int rg = threadIdx.x / tpr;    // row group within wg
  const int cbase = (threadIdx.x % tpr) << 3;
  extern __shared__ float lds[];       // [2][rows][C]
  float* s_sum = lds;
  float* s_sq = lds + (long)rows * C;

  float sum[8], sq[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) { sum[i] = 0.f; sq[i] = 0.f; }

  for (long r = (long)blockIdx.x * rows + rg; r < M;
       r += (long)gridDim.x * rows) {
    Vec8 vec;
    vec.raw = *reinterpret_cast<const uint4*>(x + r * C + cbase);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float f = bf2f(vec.v[i]);
      sum[i] += f;
      sq[i] += f * f;
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    s_sum[(long)rg * C + cbase + i] = sum[i];
    s_sq[(long)rg * C + cbase + i] = sq[i];
  }
  __syncthreads();
  // Threads 0..C-1 reduce over row groups (C <= 256 assumed; python layer
  // enforces C <= 2048 by splitting, in practice C is 64..512).
  for (int c = threadIdx.x; c < C; c += 256) {
    float a = 0.f, b = 0.f;
    for (int g = 0; g < rows; ++g) {
      a += s_sum[(long)g * C + c];
      b += s_sq[(long)g * C + c];
    }
    atomicAdd(&psum[c], a);
    atomicAdd(&psq[c], b);
  }
}

// ---------------------------------------------------------------------------
// Finalize: mean/invstd + scale/shift + running-stat update (1 workgroup).
// ---------------------------------------------------------------------------

extern "C" __global__ void bn_finalize_kernel(
    const float* __restrict__ psum, const float* __restrict__ psq,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    float* __restrict__ scale_out, float* __restrict__ shift_out,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    long M, int C, float eps, float momentum) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    float mean = psum[c] / (float)M;
    float var = fmaxf(psq[c] / (float)M - mean * mean, 0.f);
    float invstd = rsqrtf(var + eps);
    float sc = gamma[c] * invstd;
    mean_out[c] = mean;
    invstd_out[c] = invstd;
    scale_out[c] = sc;
    shift_out[c] = beta[c] - mean * sc;
    if (running_mean != nullptr) {
      // torch semantics: running stats use unbiased variance.
      float unbiased = (M > 1) ? var * (float)M / (float)(M - 1) : var;
      running_mean[c] += momentum * (mean - running_mean[c]);
      running_var[c] += momentum * (unbiased - running_var[c]);
    }
  }
}

// ---------------------------------------------------------------------------
// Apply: y = relu?(x * scale + shift), one coalesced pass.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
bn_apply_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                const float* __restrict__ scale,
                const float* __restrict__ shift, long M, int C,
                int fuse_relu) {
  const int tpr = C >> 3;
  const int rows = 256 / tpr;
  const int rg = threadIdx.x / tpr;
  const int cbase = (threadIdx.x % tpr) << 3;
  float sc[8], sh[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    sc[i] = scale[cbase + i];
    sh[i] = shift[cbase + i];
  }
  for (long r = (long)blockIdx.x * rows + rg; r < M;
       r += (long)gridDim.x * rows) {
    Vec8 in, out;
    in.raw = *reinterpret_cast<const uint4*>(x + r * C + cbase);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float f = bf2f(in.v[i]) * sc[i] + sh[i];
      if (fuse_relu) f = fmaxf(f, 0.f);
      out.v[i] = f2bf(f);
    }
    *reinterpret_cast<uint4*>(y + r * C + cbase) = out.raw;
  }
}

// ---------------------------------------------------------------------------
// Backward reduction: dbeta = sum g, dgamma = sum g*xhat, g = dy * (y>0).
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
bn_bwd_reduce_kernel(const bf16_t* __restrict__ x,
                     const bf16_t* __restrict__ dy,
                     const bf16_t* __restrict__ y,
                     const float* __restrict__ mean,
                     const float* __restrict__ invstd,
                     float* __restrict__ pdbeta,
                     float* __restrict__ pdgamma, long M, int C,
                     int fused_relu) {
  const int tpr = C >> 3;
  const int rows = 256 / tpr;
  const int rg = threadIdx.x / tpr;
  const int cbase = (threadIdx.x % tpr) << 3;
  extern __shared__ float lds[];
  float* s_db = lds;
  float* s_dg = lds + (long)rows * C;
  float mu[8], is[8], db[8], dg[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    mu[i] = mean[cbase + i];
    is[i] = invstd[cbase + i];
    db[i] = 0.f;
    dg[i] = 0.f;
  }
  for (long r = (long)blockIdx.x * rows + rg; r < M;
       r += (long)gridDim.x * rows) {
    Vec8 vx, vdy, vy;
    vx.raw = *reinterpret_cast<const uint4*>(x + r * C + cbase);
    vdy.raw = *reinterpret_cast<const uint4*>(dy + r * C + cbase);
    if (fused_relu) {
      vy.raw = *reinterpret_cast<const uint4*>(y + r * C + cbase);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float g = bf2f(vdy.v[i]);
      if (fused_relu && bf2f(vy.v[i]) <= 0.f) g = 0.f;
      float xhat = (bf2f(vx.v[i]) - mu[i]) * is[i];
      db[i] += g;
      dg[i] += g * xhat;
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    s_db[(long)rg * C + cbase + i] = db[i];
    s_dg[(long)rg * C + cbase + i] = dg[i];
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += 256) {
    float a = 0.f, b = 0.f;
    for (int g = 0; g < rows; ++g) {
      a += s_db[(long)g * C + c];
      b += s_dg[(long)g * C + c];
    }
    atomicAdd(&pdbeta[c], a);
    atomicAdd(&pdgamma[c], b);
  }
}

// ---------------------------------------------------------------------------
// Backward dx: dx = k1*g + k2*x + k3 with per-channel coefficients.
//   dx = gamma*invstd * (g - dbeta/M - xhat * dgamma/M)
// ---------------------------------------------------------------------------

extern "C" __global__ void bn_bwd_coeffs_kernel(
    const float* __restrict__ gamma, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ dbeta,
    const float* __restrict__ dgamma, float* __restrict__ k1,
    float* __restrict__ k2, float* __restrict__ k3, long M, int C) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    float gs = gamma[c] * invstd[c];
    float t = dgamma[c] / (float)M * invstd[c];
    k1[c] = gs;
    k2[c] = -gs * t;
    k3[c] = gs * (mean[c] * t - dbeta[c] / (float)M);
  }
}

extern "C" __global__ void __launch_bounds__(256)
bn_bwd_dx_kernel(const bf16_t* __restrict__ x,
                 const bf16_t* __restrict__ dy,
                 const bf16_t* __restrict__ y,
                 const float* __restrict__ k1, const float* __restrict__ k2,
                 const float* __restrict__ k3, bf16_t* __restrict__ dx,
                 long M, int C, int fused_relu) {
  const int tpr = C >> 3;
  const int rows = 256 / tpr;
  const int rg = threadIdx.x / tpr;
  const int cbase = (threadIdx.x % tpr) << 3;
  float a[8], b[8], c3[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = k1[cbase + i];
    b[i] = k2[cbase + i];
    c3[i] = k3[cbase + i];
  }
  for (long r = (long)blockIdx.x * rows + rg; r < M;
       r += (long)gridDim.x * rows) {
    Vec8 vx, vdy, vy, out;
    vx.raw = *reinterpret_cast<const uint4*>(x + r * C + cbase);
    vdy.raw = *reinterpret_cast<const uint4*>(dy + r * C + cbase);
    if (fused_relu) {
      vy.raw = *reinterpret_cast<const uint4*>(y + r * C + cbase);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float g = bf2f(vdy.v[i]);
      if (fused_relu && bf2f(vy.v[i]) <= 0.f) g = 0.f;
      float xv = bf2f(vx.v[i]);
      out.v[i] = f2bf(a[i] * g + b[i] * xv + c3[i]);
    }
    *reinterpret_cast<uint4*>(dx + r * C + cbase) = out.raw;
  }
}

// ===========================================================================
// Launchers
// ===========================================================================

static void check_flat(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be a contiguous [M,C] view");
}

static int pick_grid(long M, int rows) {
  long wgs = (M + rows - 1) / rows;
  if (wgs > 4096) wgs = 4096;  // oversubscribe 256 CUs, bounded atomics
  if (wgs < 1) wgs = 1;
  return (int)wgs;
}

std::vector<at::Tensor> fused_bn_relu_forward(
    at::Tensor x, at::Tensor gamma, at::Tensor beta,
    c10::optional<at::Tensor> running_mean,
    c10::optional<at::Tensor> running_var, double eps, double momentum,
    bool fuse_relu) {
  check_flat(x, "x");
  const long M = x.size(0);
  const int C = (int)x.size(1);
  TORCH_CHECK(C % 8 == 0 && C <= 2048, "C must be a multiple of 8, <=2048");
  auto opts = x.options().dtype(at::kFloat);
  auto psum = at::zeros({C}, opts);
  auto psq = at::zeros({C}, opts);
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  auto scale = at::empty({C}, opts);
  auto shift = at::empty({C}, opts);
  auto y = at::empty_like(x);

  const int tpr = C / 8;
  const int rows = 256 / tpr;
  const int grid = pick_grid(M, rows);
  const size_t lds_bytes = 2l * rows * C * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();

  hipLaunchKernelGGL(bn_stats_kernel, dim3(grid), dim3(256), lds_bytes,
                     stream.stream(),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     psum.data_ptr<float>(), psq.data_ptr<float>(), M, C);
  float* rm = running_mean.has_value()
                  ? running_mean->data_ptr<float>() : nullptr;
  float* rv = running_var.has_value()
                  ? running_var->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(1), dim3(256), 0,
                     stream.stream(), psum.data_ptr<float>(),
                     psq.data_ptr<float>(), gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(), rm, rv, M, C, (float)eps,
                     (float)momentum);
  hipLaunchKernelGGL(bn_apply_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     reinterpret_cast<bf16_t*>(y.data_ptr()),
                     scale.data_ptr<float>(), shift.data_ptr<float>(), M, C,
                     fuse_relu ? 1 : 0);
  return {y, mean, invstd};
}

at::Tensor bn_inference_apply(at::Tensor x, at::Tensor scale,
                              at::Tensor shift, bool fuse_relu) {
  check_flat(x, "x");
  const long M = x.size(0);
  const int C = (int)x.size(1);
  auto y = at::empty_like(x);
  const int tpr = C / 8;
  const int rows = 256 / tpr;
  const int grid = pick_grid(M, rows);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(bn_apply_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     reinterpret_cast<bf16_t*>(y.data_ptr()),
                     scale.data_ptr<float>(), shift.data_ptr<float>(), M, C,
                     fuse_relu ? 1 : 0);
  return y;
}

std::vector<at::Tensor> fused_bn_relu_backward(
    at::Tensor dy, at::Tensor x, at::Tensor y, at::Tensor gamma,
    at::Tensor mean, at::Tensor invstd, bool fused_relu) {
  check_flat(x, "x");
  check_flat(dy, "dy");
  const long M = x.size(0);
  const int C = (int)x.size(1);
  auto opts = x.options().dtype(at::kFloat);
  auto dbeta = at::zeros({C}, opts);
  auto dgamma = at::zeros({C}, opts);
  auto k1 = at::empty({C}, opts);
  auto k2 = at::empty({C}, opts);
  auto k3 = at::empty({C}, opts);
  auto dx = at::empty_like(x);
  const int tpr = C / 8;
  const int rows = 256 / tpr;
  const int grid = pick_grid(M, rows);
  const size_t lds_bytes = 2l * rows * C * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(grid), dim3(256), lds_bytes,
                     stream.stream(),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                     reinterpret_cast<const bf16_t*>(y.data_ptr()),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     dbeta.data_ptr<float>(), dgamma.data_ptr<float>(), M, C,
                     fused_relu ? 1 : 0);
  hipLaunchKernelGGL(bn_bwd_coeffs_kernel, dim3(1), dim3(256), 0,
                     stream.stream(), gamma.data_ptr<float>(),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     dbeta.data_ptr<float>(), dgamma.data_ptr<float>(),
                     k1.data_ptr<float>(), k2.data_ptr<float>(),
                     k3.data_ptr<float>(), M, C);
  hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                     reinterpret_cast<const bf16_t*>(y.data_ptr()),
                     k1.data_ptr<float>(), k2.data_ptr<float>(),
                     k3.data_ptr<float>(),
                     reinterpret_cast<bf16_t*>(dx.data_ptr()), M, C,
                     fused_relu ? 1 : 0);
  return {dx, dgamma, dbeta};
}
