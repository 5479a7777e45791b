// im2col / col2im for NHWC bf16 convs (gfx950).
//
// The ResNet-family conv path (tensor2robot_amd/ops/gemm_conv.py):
// MIOpen on these shapes dispatches thousands of SubTensorOp/fill
// helper kernels per step and parks in find for minutes (profiles/
// r2: BC-Z FiLM-ResNet18 at 21 ms/step, ~2700 helper dispatches per
// step).  The MI355X-first shape for them is ONE gather kernel + ONE
// library GEMM (rocBLAS bf16, MFMA-saturating) per direction:
//
//   fwd : col = im2col(x)                [M, RS*C]
//         y   = col @ w_mat              [M, K]      (rocBLAS)
//   dw  : dw_mat = col^T @ dy            [RS*C, K]   (rocBLAS)
//   dx  : dcol = dy @ w_mat^T            [M, RS*C]   (rocBLAS)
//         dx   = col2im_gather(dcol)     per-input-pixel gather, no
//                                        atomics.
//
// Both kernels move 16 B per lane over the channel dim (C % 8 == 0),
// coalesced on read and write.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 i2c_bf16;

// col[m][rs*C + c] = x[n, ih, iw, c] with (ih, iw) = stride*(oh, ow) +
// (r, s) - pad; m = ((n*OH)+oh)*OW + ow.  One thread = one 8-channel
// chunk of one (m, rs) cell.
extern "C" __global__ void __launch_bounds__(256)
im2col_nhwc_kernel(const i2c_bf16* __restrict__ x,
                   i2c_bf16* __restrict__ col,
                   int N, int H, int W, int C,
                   int R, int S, int pad, int stride,
                   int OH, int OW) {
  const int c8 = C >> 3;
  const long cells = (long)N * OH * OW * R * S * c8;
  const int RSC8 = R * S * c8;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int chunk = (int)(i % RSC8);
    const long m = i / RSC8;
    const int cc = chunk % c8;
    const int rs = chunk / c8;
    const int r = rs / S, s = rs % S;
    const int ow = (int)(m % OW);
    const int oh = (int)((m / OW) % OH);
    const int n = (int)(m / ((long)OH * OW));
    const int ih = oh * stride + r - pad;
    const int iw = ow * stride + s - pad;
    uint4 v = make_uint4(0, 0, 0, 0);
    if (ih >= 0 && ih < H && iw >= 0 && iw < W) {
      v = *reinterpret_cast<const uint4*>(
          x + (((long)n * H + ih) * W + iw) * C + cc * 8);
    }
    *reinterpret_cast<uint4*>(
        col + m * ((long)R * S * C) + (long)rs * C + cc * 8) = v;
  }
}

// dx[n, ih, iw, c] = sum over (r, s) of dcol[m(oh, ow)][rs*C + c]
// where oh = (ih + pad - r) / stride when divisible and in range.
// Gather form: no atomics, each input chunk written once.
extern "C" __global__ void __launch_bounds__(256)
col2im_nhwc_kernel(const i2c_bf16* __restrict__ dcol,
                   i2c_bf16* __restrict__ dx,
                   int N, int H, int W, int C,
                   int R, int S, int pad, int stride,
                   int OH, int OW) {
  const int c8 = C >> 3;
  const long cells = (long)N * H * W * c8;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int cc = (int)(i % c8);
    const long pix = i / c8;
    const int iw = (int)(pix % W);
    const int ih = (int)((pix / W) % H);
    const int n = (int)(pix / ((long)H * W));
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
    for (int r = 0; r < R; ++r) {
      const int oh_num = ih + pad - r;
      if (oh_num < 0 || oh_num % stride) continue;
      const int oh = oh_num / stride;
      if (oh >= OH) continue;
      for (int s = 0; s < S; ++s) {
        const int ow_num = iw + pad - s;
        if (ow_num < 0 || ow_num % stride) continue;
        const int ow = ow_num / stride;
        if (ow >= OW) continue;
        const long m = ((long)n * OH + oh) * OW + ow;
        const i2c_bf16* src = dcol + m * ((long)R * S * C)
            + ((long)r * S + s) * C + cc * 8;
        uint4 v = *reinterpret_cast<const uint4*>(src);
        const i2c_bf16* sv = reinterpret_cast<const i2c_bf16*>(&v);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += __bfloat162float(sv[j]);
      }
    }
    uint4 out;
    i2c_bf16* ov = reinterpret_cast<i2c_bf16*>(&out);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = __float2bfloat16(acc[j]);
    *reinterpret_cast<uint4*>(
        dx + (((long)n * H + ih) * W + iw) * C + cc * 8) = out;
  }
}

static int i2c_grid(long cells) {
  long wgs = (cells + 255) / 256;
  if (wgs > 4096) wgs = 4096;
  if (wgs < 1) wgs = 1;
  return (int)wgs;
}

at::Tensor im2col_nhwc(at::Tensor x, int64_t R, int64_t S, int64_t pad,
                       int64_t stride) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "im2col_nhwc: bf16 CUDA input required");
  x = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "im2col_nhwc: C % 8 == 0");
  const int OH = (int)((H + 2 * pad - R) / stride + 1);
  const int OW = (int)((W + 2 * pad - S) / stride + 1);
  auto col = at::empty({(long)N * OH * OW, R * S * C}, x.options());
  const long cells = (long)N * OH * OW * R * S * (C / 8);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(im2col_nhwc_kernel, dim3(i2c_grid(cells)),
                     dim3(256), 0, stream.stream(),
                     (const i2c_bf16*)x.data_ptr(),
                     (i2c_bf16*)col.data_ptr(),
                     N, H, W, C, (int)R, (int)S, (int)pad, (int)stride,
                     OH, OW);
  return col;
}

at::Tensor col2im_nhwc(at::Tensor dcol, int64_t N, int64_t C, int64_t H,
                       int64_t W, int64_t R, int64_t S, int64_t pad,
                       int64_t stride) {
  TORCH_CHECK(dcol.is_cuda() && dcol.scalar_type() == at::kBFloat16,
              "col2im_nhwc: bf16 CUDA input required");
  dcol = dcol.contiguous();
  const int OH = (int)((H + 2 * pad - R) / stride + 1);
  const int OW = (int)((W + 2 * pad - S) / stride + 1);
  TORCH_CHECK(dcol.size(0) == (long)N * OH * OW &&
              dcol.size(1) == R * S * C, "col2im_nhwc: dcol shape");
  auto dx = at::empty({N, C, H, W}, dcol.options(),
                      at::MemoryFormat::ChannelsLast);
  const long cells = (long)N * H * W * (C / 8);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(col2im_nhwc_kernel, dim3(i2c_grid(cells)),
                     dim3(256), 0, stream.stream(),
                     (const i2c_bf16*)dcol.data_ptr(),
                     (i2c_bf16*)dx.data_ptr(),
                     (int)N, (int)H, (int)W, (int)C, (int)R, (int)S,
                     (int)pad, (int)stride, OH, OW);
  return dx;
}
