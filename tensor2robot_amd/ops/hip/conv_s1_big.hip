// C-chunked MFMA conv for big-channel stride-1 3x3 NHWC bf16 (gfx950).
//
// Covers the ResNet-family 3x3 bottleneck/basic convs (C=K in
// {128,256,512}; reference film_resnet_model.py:100-341,
// grasp2vec/resnet.py towers at 472^2) where the GEMM-conv path
// (im2col + rocBLAS) pays 9x gather traffic and the col GEMM caps at
// ~225 TF (profiles/r2_bcz_gemm_conv.md).  Backward-data is the same
// kernel on flipped/transposed prepacked weights (conv.py).
//
// Design (cdna_hip_programming.md §2/§3/§5, T14):
//  * implicit GEMM like conv_s1.hip: a 256-thread WG computes an
//    8x16-pixel tile x 64 output channels; 4 waves x (32 pixels x 64).
//  * C is processed in 32-channel chunks: each chunk stages the
//    10x18-pixel x-halo slice (PITCH 40 shorts = 80 B = five 16-B
//    slots, an odd multiple -> the four 16-lane ds_read_b128 groups
//    hit 16 distinct slots each: conflict-free, same argument as the
//    proven 144-B pitch at 64 ch) and the chunk's 9*2 weight rows
//    ([rs][c16][n(64)][16], 32-B n-stride) into ONE LDS buffer
//    (51.3 KB -> 2 WGs/CU), then runs the 36-MFMA chunk loop between
//    two barriers.
//  * staging is direct global->LDS with hoisted per-thread addresses;
//    the two resident WGs per CU desynchronize and cover each other's
//    staging latency.  (A register-held T14 prefetch was tried: the
//    compiler spills the 12 uint4 to scratch at every VGPR budget.)
//  * fragment maps identical to conv_s1.hip (verified by mfma_probe):
//    A row = l%32, k = (l>>5)*8+j; B col = l%32;
//    C/D col = l&31, row = (reg&3)+8*(reg>>2)+4*(l>>5).
//
// Weight image: the SAME pack_conv_w output as conv_s1.hip
// ([rs][c16][n(K)][24]); chunks address c16 slices of it directly.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 bbf16_t;
typedef __attribute__((ext_vector_type(8))) short bbf16x8;
typedef __attribute__((ext_vector_type(16))) float bf32x16;

#define BT_H 8
#define BT_W 16
#define BHALO_H (BT_H + 2)     // R = 3
#define BHALO_W (BT_W + 2)
#define BXP 40                 // shorts per pixel row: 32 ch + 8 pad
#define BWPAD 24               // global packed-image n-row stride

#define BXTILE (BHALO_H * BHALO_W * BXP)          // 7200 shorts
#define BWALL (9 * 2 * 64 * 16)                   // 18432 shorts
#define BXCHUNKS (BHALO_H * BHALO_W * 4)          // 16-B pieces: 720
#define BWHALVES (9 * 2 * 64 * 2)                 // 16-B pieces: 2304

// Per-thread register staging counts (256 threads).
#define BXR 3                                     // ceil(720/256)
#define BWR 9                                     // 2304/256

__global__ void __launch_bounds__(256, 2)
conv_s1_nhwc_cchunk_kernel(const bbf16_t* __restrict__ x,
                           const bbf16_t* __restrict__ wpk,
                           bbf16_t* __restrict__ y,
                           int N, int H, int W, int C, int K, int pad,
                           int OH, int OW, int tiles_h, int tiles_w) {
  __shared__ short lds[BXTILE + BWALL];
  short* xtile = lds;
  short* wall = lds + BXTILE;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int n0 = blockIdx.y * 64;
  const int c16n_full = C >> 4;
  const int chunks = C >> 5;

  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int th = trest / tiles_w;
  const int tw = trest % tiles_w;
  const int oh0 = th * BT_H, ow0 = tw * BT_W;

  // Per-thread staging addresses are chunk-invariant except for the
  // channel offset (x: += 32 bf16) / c16 offset (w: += 2 rows): hoist
  // them; the staging bodies are then one load + one LDS store each.
  // (A register-held next-chunk prefetch was tried first: the compiler
  // spills the 12 uint4 to scratch at every VGPR budget — direct
  // staging + 2 resident WGs/CU hides the latency instead.)
  const int x_ci[BXR] = {tid, tid + 256, tid + 512};
  const bbf16_t* x_src[BXR];
  short* x_dst[BXR];
#pragma unroll
  for (int i = 0; i < BXR; ++i) {
    const int ci = x_ci[i];
    const int ch8 = ci & 3;
    const int pix = ci >> 2;
    const int hrow = pix / BHALO_W, hcol = pix % BHALO_W;
    const int iy = oh0 - pad + hrow;
    const int ix = ow0 - pad + hcol;
    const bool ok = ci < BXCHUNKS && iy >= 0 && iy < H && ix >= 0 &&
        ix < W;
    x_src[i] = ok ? x + (((long)img * H + iy) * W + ix) * C + ch8 * 8
                  : nullptr;
    x_dst[i] = (ci < BXCHUNKS) ? &xtile[pix * BXP + ch8 * 8] : nullptr;
  }
  // w staging addresses: piece i handles nrow = (tid + i*256) >> 1;
  // the global source advances by 2*K*BWPAD per chunk (c16g += 2).
  const bbf16_t* w_src[BWR];
  short* w_dst[BWR];
#pragma unroll
  for (int i = 0; i < BWR; ++i) {
    const int hi = tid + i * 256;                // < 2304 always
    const int nrow = hi >> 1, half = (hi & 1) * 8;
    const int nn = nrow & 63;
    const int rcl = nrow >> 6;                   // rs*2 + c16l, < 18
    const int rs = rcl >> 1;
    w_src[i] = &wpk[((long)(rs * c16n_full + (rcl & 1)) * K + n0 + nn)
                    * BWPAD + half];
    w_dst[i] = &wall[nrow * 16 + half];
  }
  const long w_chunk_stride = 2L * K * BWPAD;    // c16g += 2

  bf32x16 acc[2];
  acc[0] = (bf32x16){};
  acc[1] = (bf32x16){};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
  const int prow = (wave * 32 + mrow) / BT_W;
  const int pcol = (wave * 32 + mrow) % BT_W;

  for (int cc = 0; cc < chunks; ++cc) {
    if (cc) __syncthreads();   // prior chunk's reads done: LDS writable
    // Direct global -> LDS staging (independent pieces: the compiler
    // batches the loads ahead of the stores).
#pragma unroll
    for (int i = 0; i < BXR; ++i) {
      uint4 v = make_uint4(0, 0, 0, 0);
      if (x_src[i])
        v = *reinterpret_cast<const uint4*>(x_src[i] + cc * 32);
      if (x_dst[i]) *reinterpret_cast<uint4*>(x_dst[i]) = v;
    }
#pragma unroll
    for (int i = 0; i < BWR; ++i)
      *reinterpret_cast<uint4*>(w_dst[i]) =
          *reinterpret_cast<const uint4*>(w_src[i]
                                          + cc * w_chunk_stride);
    __syncthreads();           // staging visible
#pragma unroll
    for (int rs = 0; rs < 9; ++rs) {
      const int r = rs / 3, s = rs % 3;
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        bbf16x8 a_frag = *reinterpret_cast<const bbf16x8*>(
            &xtile[((prow + r) * BHALO_W + (pcol + s)) * BXP
                   + c16 * 16 + kgrp * 8]);
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          bbf16x8 b_frag = *reinterpret_cast<const bbf16x8*>(
              &wall[((rs * 2 + c16) * 64 + nt * 32 + mrow) * 16
                    + kgrp * 8]);
          acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_frag, b_frag, acc[nt], 0, 0, 0);
        }
      }
    }
  }

  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / BT_W;
      const int ocol = ow0 + p % BT_W;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * K
          + n0 + nt * 32 + ocol_n] = __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}

at::Tensor conv_s1_nhwc_cchunk(at::Tensor x, at::Tensor wpk, int64_t K,
                               int64_t R, int64_t S, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "cchunk: 4D CUDA input");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "cchunk: bf16 input");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "cchunk: channels_last input");
  TORCH_CHECK(R == 3 && S == 3, "cchunk: 3x3 only");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 32 == 0 && C >= 32, "cchunk: C % 32");
  TORCH_CHECK(K % 64 == 0, "cchunk: K % 64");
  const int OH = H + 2 * (int)pad - 2;
  const int OW = W + 2 * (int)pad - 2;
  TORCH_CHECK(OH > 0 && OW > 0, "cchunk: empty output");
  auto y = at::empty({N, K, OH, OW},
                     x.options().memory_format(
                         at::MemoryFormat::ChannelsLast));
  const int tiles_h = (OH + BT_H - 1) / BT_H;
  const int tiles_w = (OW + BT_W - 1) / BT_W;
  const long grid_x = (long)N * tiles_h * tiles_w;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(conv_s1_nhwc_cchunk_kernel,
                     dim3(grid_x, K / 64), dim3(256), 0,
                     stream.stream(),
                     (const bbf16_t*)x.data_ptr(),
                     (const bbf16_t*)wpk.data_ptr(),
                     (bbf16_t*)y.data_ptr(),
                     N, H, W, C, (int)K, (int)pad, OH, OW,
                     tiles_h, tiles_w);
  return y;
}
