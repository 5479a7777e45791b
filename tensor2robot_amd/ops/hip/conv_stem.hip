// MFMA stem conv: 6x6 stride-2 pad-2, C=3 -> K=64, NHWC bf16 (gfx950).
//
// The Grasping44 stem (472^2x3 -> 236^2x64) is pathological for
// MIOpen's igemm (C=3 gives a K'=108 GEMM it runs at <5% MFMA,
// ~0.23 ms/step).  Being the FIRST layer, only the forward matters
// (the input carries no gradient); wrw stays on MIOpen.
//
// Implicit im2col: K' = r(6) x [s(6) x c(3) = 18, padded to 32] = 192.
// For one output pixel and fixed r, the 18 taps x[iy+r][ix0..ix0+5][0..2]
// are CONTIGUOUS in NHWC memory, so each (pixel, r) row of the A-image
// is built with three overlapping 16-B loads (edge pixels fall back to
// scalar taps).  Zero padding in the 18->32 slots contributes nothing.
// A-image [128 pixels][200 (192 + bank pad)], B prepacked host-side to
// [kstep(12)][n(64)][24 (16 + pad)] — both fragment reads are
// contiguous ds_read_b128 with conflict-free strides.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 sbf16_t;
typedef __attribute__((ext_vector_type(8))) short sbf16x8;
typedef __attribute__((ext_vector_type(16))) float sf32x16;

#define ST_R 6
#define ST_S 6
#define ST_C 3
#define ST_K 64
#define ST_STRIDE 2
#define ST_PAD 2
#define ST_TILE_H 8
#define ST_TILE_W 16
#define ST_PIX (ST_TILE_H * ST_TILE_W)     // 128
#define ST_KP 192                           // 6 r-chunks x 32
#define ST_APITCH 200                       // + bank pad
#define ST_WROW 24                          // 16 + pad per n-row
#define ST_KSTEPS (ST_KP / 16)              // 12

__global__ void __launch_bounds__(256, 1)
conv_stem_kernel(const sbf16_t* __restrict__ x,
                 const sbf16_t* __restrict__ wpk,  // [12][64][24] bf16
                 sbf16_t* __restrict__ y,
                 int N, int H, int W, int OH, int OW,
                 int tiles_h, int tiles_w) {
  __shared__ short lds_a[ST_PIX * ST_APITCH];         // 51.2 KiB
  __shared__ short lds_b[ST_KSTEPS * ST_K * ST_WROW]; // 36.9 KiB

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int oh0 = (trest / tiles_w) * ST_TILE_H;
  const int ow0 = (trest % tiles_w) * ST_TILE_W;

  // ---- stage B once (layout matches global bytes) ----
  {
    constexpr int pieces = (ST_KSTEPS * ST_K * ST_WROW * 2) / 16;
    for (int i = tid; i < pieces; i += 256) {
      *reinterpret_cast<uint4*>(&lds_b[i * 8]) =
          *reinterpret_cast<const uint4*>(&wpk[i * 8]);
    }
  }

  // ---- build the A-image: one (pixel, r) row per work item ----
  {
    for (int i = tid; i < ST_PIX * ST_R; i += 256) {
      const int p = i / ST_R, r = i % ST_R;
      const int prow = p / ST_TILE_W, pcol = p % ST_TILE_W;
      const int iy = (oh0 + prow) * ST_STRIDE - ST_PAD + r;
      const int ix0 = (ow0 + pcol) * ST_STRIDE - ST_PAD;
      short* dst = &lds_a[p * ST_APITCH + r * 32];
      const bool row_ok = iy >= 0 && iy < H;
      if (row_ok && ix0 >= 0 && ix0 + ST_S <= W) {
        // Interior: 18 contiguous taps; copy 24 elements (the extra 6
        // land in the zero-pad slots and are overwritten below).
        const sbf16_t* src = x + (((long)img * H + iy) * W + ix0) * ST_C;
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          *reinterpret_cast<uint4*>(dst + j * 8) =
              *reinterpret_cast<const uint4*>(
                  reinterpret_cast<const short*>(src) + j * 8);
        }
#pragma unroll
        for (int j = 18; j < 32; ++j) dst[j] = 0;
      } else {
#pragma unroll
        for (int s = 0; s < ST_S; ++s) {
          const int ix = ix0 + s;
          const bool ok = row_ok && ix >= 0 && ix < W;
          const sbf16_t* src =
              x + (((long)img * H + iy) * W + ix) * ST_C;
#pragma unroll
          for (int c = 0; c < ST_C; ++c) {
            dst[s * ST_C + c] =
                ok ? *reinterpret_cast<const short*>(src + c) : 0;
          }
        }
#pragma unroll
        for (int j = 18; j < 32; ++j) dst[j] = 0;
      }
    }
  }
  __syncthreads();

  // ---- 12 ksteps x 2 n-tiles ----
  sf32x16 acc[2] = {{}, {}};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
#pragma unroll
  for (int kstep = 0; kstep < ST_KSTEPS; ++kstep) {
    sbf16x8 a_frag = *reinterpret_cast<const sbf16x8*>(
        &lds_a[(wave * 32 + mrow) * ST_APITCH + kstep * 16 + kgrp * 8]);
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int n = nt * 32 + mrow;
      sbf16x8 b_frag = *reinterpret_cast<const sbf16x8*>(
          &lds_b[(kstep * ST_K + n) * ST_WROW + kgrp * 8]);
      acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          a_frag, b_frag, acc[nt], 0, 0, 0);
    }
  }

  // ---- epilogue ----
  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / ST_TILE_W;
      const int ocol = ow0 + p % ST_TILE_W;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * ST_K
          + nt * 32 + ocol_n] = __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}

at::Tensor conv_stem_nhwc(at::Tensor x, at::Tensor wpk) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_stem: bf16 CUDA input required");
  TORCH_CHECK(x.dim() == 4 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_stem: channels_last required");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C == 3, "conv_stem: C == 3 only");
  const int OH = (H + 2 * ST_PAD - ST_R) / ST_STRIDE + 1;
  const int OW = (W + 2 * ST_PAD - ST_S) / ST_STRIDE + 1;
  auto y = at::empty({N, ST_K, OH, OW},
                     x.options().memory_format(
                         at::MemoryFormat::ChannelsLast));
  const int tiles_h = (OH + ST_TILE_H - 1) / ST_TILE_H;
  const int tiles_w = (OW + ST_TILE_W - 1) / ST_TILE_W;
  const long grid = (long)N * tiles_h * tiles_w;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(conv_stem_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     (const sbf16_t*)x.data_ptr(),
                     (const sbf16_t*)wpk.data_ptr(),
                     (sbf16_t*)y.data_ptr(),
                     N, H, W, OH, OW, tiles_h, tiles_w);
  return y;
}

// Host-side weight pack for the stem: w [64, 3, 6, 6] ->
// [kstep(12)][n(64)][24] where k' = r*32 + (s*3 + c), zeros elsewhere.
template <int DUMMY>
__global__ void __launch_bounds__(256)
pack_stem_w_kernel(const sbf16_t* __restrict__ w,   // [K][C][R][S]
                   sbf16_t* __restrict__ out) {
  const long total = (long)ST_KSTEPS * ST_K * ST_WROW;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int cc = i % ST_WROW;
    long rest = i / ST_WROW;
    const int n = rest % ST_K;
    const int kstep = rest / ST_K;
    sbf16_t v = __float2bfloat16(0.0f);
    if (cc < 16) {
      const int kprime = kstep * 16 + cc;     // within [0, 192)
      const int r = kprime / 32;
      const int sc = kprime % 32;
      if (sc < ST_S * ST_C) {
        const int s = sc / ST_C, c = sc % ST_C;
        v = w[(((long)n * ST_C + c) * ST_R + r) * ST_S + s];
      }
    }
    out[i] = v;
  }
}

at::Tensor pack_stem_w(at::Tensor w) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && w.size(0) == ST_K &&
              w.size(1) == ST_C && w.size(2) == ST_R &&
              w.size(3) == ST_S,
              "pack_stem_w: [64, 3, 6, 6] required");
  w = w.contiguous();
  if (w.scalar_type() != at::kBFloat16) w = w.to(at::kBFloat16);
  auto out = at::empty({ST_KSTEPS, ST_K, ST_WROW}, w.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pack_stem_w_kernel<0>, dim3(72), dim3(256), 0,
                     stream.stream(), (const sbf16_t*)w.data_ptr(),
                     (sbf16_t*)out.data_ptr());
  return out;
}
