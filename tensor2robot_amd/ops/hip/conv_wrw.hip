// MFMA weight-gradient (wrw) kernel for stride-1 NHWC bf16 convs.
//
// dW[rs][c][k] = sum over pixels p of x[p + D(rs), c] * dy[p, k] — a
// GEMM with M = C, N = K and the 200k-pixel axis as the contraction.
// Both operands are channel-minor in memory while the MFMA fragments
// want PIXEL-minor rows, so each spatial window's x-halo and dy tiles
// are staged TRANSPOSED in LDS ([c][p] / [k][p] images, padded pixel
// strides for conflict-free ds_read_b128).
//
// Accumulation strategy (the naive K-split needs 163M atomics/step —
// rejected): a workgroup owns an RS-GROUP of up to 5 (r,s) offsets and
// accumulates its dW slice in an 80-KiB LDS array across MANY spatial
// windows; one atomicAdd pass per WG at the end (≈6.5M atomics total).
// Each (rs, c, k) cell is read-modify-written by exactly one lane, so
// the in-LDS accumulation is race-free.
//
// LDS budget (C=K=64): xT 64x248x2 = 31.7 KiB + dyT 64x136x2 = 17.4 KiB
// + dW 5x64x64x4 = 80 KiB = 129 KiB -> one 256-thread WG per CU.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 wbf16_t;
typedef __attribute__((ext_vector_type(8))) short wbf16x8;
typedef __attribute__((ext_vector_type(16))) float wf32x16;

#define WTILE_H 8
#define WTILE_W 16
#define WWIN_P (WTILE_H * WTILE_W)        // 128 window pixels
#define WHALO_H (WTILE_H + 4)
#define WHALO_W (WTILE_W + 4)
#define WHALO_P (WHALO_H * WHALO_W)       // 240
#define XT_PITCH 248                      // halo pixels + pad (bank spread)
#define DYT_PITCH 136                     // window pixels + pad
#define RS_GROUP 5

template <int C16N, int NTILES>
__global__ void __launch_bounds__(256, 1)
conv_s1_wrw_kernel(const wbf16_t* __restrict__ x,
                   const wbf16_t* __restrict__ dy,
                   float* __restrict__ dw,     // [RS][C][K] f32, zeroed
                   int N, int H, int W, int K,
                   int R, int S, int pad,
                   int OH, int OW, int tiles_h, int tiles_w,
                   int window_groups) {
  constexpr int C = C16N * 16;
  __shared__ short lds_xt[C16N * 16 * XT_PITCH];
  __shared__ short lds_dyt[64 * DYT_PITCH];
  __shared__ float lds_dw[RS_GROUP * 64 * 64];
  // NOTE: three __shared__ objects are fine here — no glds in this
  // kernel (the §5 trap is glds-pipeline-specific).

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  // Distinct (c-tile, k-tile) quadrant per wave; waves beyond the
  // quadrant count only help with staging (C=K=64 uses all four).
  constexpr int N_MTILES = (C16N * 16) / 32;
  const bool active = wave < N_MTILES * NTILES;
  const int mtile = active ? wave % N_MTILES : 0;
  const int ntile = active ? wave / N_MTILES : 0;
  const int rs0 = blockIdx.y * RS_GROUP;
  const int RS = R * S;
  const int rs_in_group = min(RS - rs0, RS_GROUP);

  // Zero the LDS dW accumulator.
  for (int i = tid; i < RS_GROUP * 64 * 64; i += 256) lds_dw[i] = 0.0f;
  __syncthreads();

  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;

  for (int win = blockIdx.x; win < total_windows; win += window_groups) {
    const int img = win / (tiles_h * tiles_w);
    const int trest = win % (tiles_h * tiles_w);
    const int oh0 = (trest / tiles_w) * WTILE_H;
    const int ow0 = (trest % tiles_w) * WTILE_W;

    // ---- stage x halo TRANSPOSED: lds_xt[c][halo_p] ----
    {
      const int halo_h = WTILE_H + R - 1, halo_w = WTILE_W + S - 1;
      constexpr int chunks = C >> 3;
      const int total = WHALO_P * chunks;  // full capacity; zero extras
      for (int i = tid; i < total; i += 256) {
        const int chunk = i % chunks;
        const int p = i / chunks;
        const int hrow = p / WHALO_W, hcol = p % WHALO_W;
        const int iy = oh0 - pad + hrow;
        const int ix = ow0 - pad + hcol;
        wbf16x8 v = {};
        if (hrow < halo_h && hcol < halo_w &&
            iy >= 0 && iy < H && ix >= 0 && ix < W) {
          v = *reinterpret_cast<const wbf16x8*>(
              x + (((long)img * H + iy) * W + ix) * C + chunk * 8);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_xt[(chunk * 8 + j) * XT_PITCH + p] = v[j];
      }
    }
    // ---- stage dy window TRANSPOSED: lds_dyt[k][win_p] (zero OOB) ----
    {
      constexpr int chunks = 64 >> 3;  // K<=64; extra rows just unused
      const int kchunks = K >> 3;
      const int total = WWIN_P * kchunks;
      for (int i = tid; i < total; i += 256) {
        const int chunk = i % kchunks;
        const int p = i / kchunks;
        const int orow = oh0 + p / WTILE_W;
        const int ocol = ow0 + p % WTILE_W;
        wbf16x8 v = {};
        if (orow < OH && ocol < OW) {
          v = *reinterpret_cast<const wbf16x8*>(
              dy + (((long)img * OH + orow) * OW + ocol) * K + chunk * 8);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_dyt[(chunk * 8 + j) * DYT_PITCH + p] = v[j];
      }
      (void)chunks;
    }
    __syncthreads();

    // ---- accumulate this window's contribution for the rs group ----
    for (int g = 0; active && g < rs_in_group; ++g) {
      const int rs = rs0 + g;
      const int r = rs / S, s = rs % S;
      wf32x16 acc = {};
#pragma unroll
      for (int kstep = 0; kstep < WWIN_P / 16; ++kstep) {
        // The 16 contraction pixels = window row (kstep*16..+15); the
        // shifted halo indices are contiguous within the row.
        const int wrow = (kstep * 16) / WTILE_W;
        const int halo_base = (wrow + r) * WHALO_W + s;
        const int c = mtile * 32 + mrow;
        wbf16x8 a_frag = *reinterpret_cast<const wbf16x8*>(
            &lds_xt[c * XT_PITCH + halo_base + kgrp * 8]);
        const int k = ntile * 32 + mrow;
        wbf16x8 b_frag = *reinterpret_cast<const wbf16x8*>(
            &lds_dyt[k * DYT_PITCH + kstep * 16 + kgrp * 8]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag,
                                                      acc, 0, 0, 0);
      }
      // LDS RMW: each (c, k) cell belongs to exactly one lane.
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int c = mtile * 32
            + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int k = ntile * 32 + (lane & 31);
        lds_dw[(g * 64 + c) * 64 + k] += acc[reg];
      }
    }
    __syncthreads();   // everyone done before restaging the tiles
  }

  // ---- flush: one atomic pass per WG ----
  for (int i = tid; i < rs_in_group * 64 * 64; i += 256) {
    const int g = i / (64 * 64);
    const int rest = i % (64 * 64);
    const int c = rest / 64, k = rest % 64;
    if (c < C && k < K) {
      atomicAdd(&dw[((long)(rs0 + g) * C + c) * K + k],
                lds_dw[(g * 64 + c) * 64 + k]);
    }
  }
}

at::Tensor conv_s1_wrw(at::Tensor x, at::Tensor dy, int64_t R, int64_t S,
                       int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_s1_wrw: bf16 x required");
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "conv_s1_wrw: bf16 dy required");
  x = x.contiguous(at::MemoryFormat::ChannelsLast);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(C % 16 == 0 && C <= 64 && K % 32 == 0 && K <= 64,
              "conv_s1_wrw: C%16, C<=64, K%32, K<=64");
  TORCH_CHECK(OH == H + 2 * pad - R + 1 && OW == W + 2 * pad - S + 1,
              "conv_s1_wrw: dy shape mismatch");
  auto dw = at::zeros({(long)R * S, C, K},
                      x.options().dtype(at::kFloat));
  const int tiles_h = (OH + WTILE_H - 1) / WTILE_H;
  const int tiles_w = (OW + WTILE_W - 1) / WTILE_W;
  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  // Enough WGs to fill the chip several times, few enough that the
  // atomic flush stays small.
  const int window_groups = std::min(total_windows, 512);
  const int rs_groups = (int)((R * S + RS_GROUP - 1) / RS_GROUP);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(window_groups, rs_groups), dim3(256),
                       0, stream.stream(),
                       (const wbf16_t*)x.data_ptr(),
                       (const wbf16_t*)dy.data_ptr(),
                       (float*)dw.data_ptr(),
                       N, H, W, K, (int)R, (int)S, (int)pad,
                       OH, OW, tiles_h, tiles_w, window_groups);
  };
  // C must be a multiple of 32 here (full MFMA row tiles).
  if (C == 64 && K == 64) launch(conv_s1_wrw_kernel<4, 2>);
  else if (C == 32 && K == 32) launch(conv_s1_wrw_kernel<2, 1>);
  else if (C == 32 && K == 64) launch(conv_s1_wrw_kernel<2, 2>);
  else TORCH_CHECK(false, "conv_s1_wrw: unsupported C/K combo");
  return dw;
}
