// Hand-written MFMA conv for stride-1 NHWC bf16 (gfx950 / CDNA4).
//
// Covers the Grasping44 hot convs (5x5 SAME @79^2, 3x3 SAME @27^2,
// 3x3 VALID, C=K=64) where MIOpen's igemm runs at ~15-20% MFMA
// utilization on these small-channel shapes (profiles/).  Backward-data
// is the same kernel on flipped/transposed prepacked weights.
//
// Design (cdna_hip_programming.md §3/§5, MI355X_MICROARCH.md):
//  * implicit GEMM: out[p, k] = sum_{r,s,c} x[p+Δ(r,s), c] * w[r,s,c,k]
//    computed with v_mfma_f32_32x32x16_bf16; M = pixels, N = K, K' = C.
//  * one 256-thread WG computes an 8x16-pixel output tile for all K:
//    4 waves x (32 pixels x K).  The 12x20-pixel input halo tile is
//    staged in LDS once per WG with a PADDED 144-B pixel stride so the
//    A-fragment ds_read_b128 lane groups land on 16 distinct banks
//    (linear 128-B stride = 16-way conflict, the §5 GEMM trap).
//  * weights are HOST-PREPACKED to [rs][c16][n][24] (8-slot pad => 48-B
//    n-stride, conflict-free B-fragment reads) and staged per (r,s)
//    via global_load_lds (lane-linear dest, layouts match exactly),
//    double-buffered so chunk rs+1 streams while rs computes.
//  * fragment layouts verified by the mfma_probe GPU test:
//    A row = l%32, k = (l>>5)*8+j; B col = l%32 (same k map);
//    C/D col = l&31, row = (reg&3)+8*(reg>>2)+4*(l>>5).
//  * all LDS in ONE __shared__ array (a second object forces vmcnt(0)
//    before every ds_read of a glds pipeline — §5 trap 4a).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 cbf16_t;
typedef __attribute__((ext_vector_type(8))) short cbf16x8;
typedef __attribute__((ext_vector_type(16))) float cf32x16;

#define TILE_H 8
#define TILE_W 16
#define HALO_H (TILE_H + 4)   // supports R <= 5
#define HALO_W (TILE_W + 4)
#define XPITCH 72             // bf16 elements per pixel row in LDS (64+8 pad)
#define WPAD 24               // bf16 per n-row in the weight image (16+8 pad)

// LDS: x tile + 2 weight buffers, one shared object.
// x: HALO_H*HALO_W pixels * XPITCH bf16 = 240*72*2 = 34560 B
// w: 2 * (C16MAX=4) * 64 * WPAD * 2 = 2*12288 B
#define XTILE_BF16 (HALO_H * HALO_W * XPITCH)
#define WBUF_BF16 (4 * 64 * WPAD)

extern "C" __global__ void __launch_bounds__(256, 2)
conv_s1_nhwc_kernel(const cbf16_t* __restrict__ x,
                    const cbf16_t* __restrict__ wpk,
                    cbf16_t* __restrict__ y,
                    int N, int C, int H, int W,
                    int K, int R, int S, int pad,
                    int OH, int OW, int tiles_h, int tiles_w) {
  __shared__ short lds[XTILE_BF16 + 2 * WBUF_BF16];
  short* xtile = lds;
  // NOTE: no array-of-LDS-pointers (hipcc cannot statically initialize
  // addrspace(3) casts) — compute buffer bases by index.
  auto wbuf = [&](int b) -> short* {
    return lds + XTILE_BF16 + b * WBUF_BF16;
  };

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int c16n = C >> 4;            // K'-chunks of 16
  const int ntiles = K >> 5;          // 32-wide N tiles (<=2 supported)
  const int wchunk_bf16 = c16n * K * WPAD;

  // Tile coordinates.
  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int th = trest / tiles_w;
  const int tw = trest % tiles_w;
  const int oh0 = th * TILE_H, ow0 = tw * TILE_W;

  // ---- stage the x halo tile (zero OOB), 16-B chunks ----
  {
    const int halo_h = TILE_H + R - 1, halo_w = TILE_W + S - 1;
    const int chunks = C >> 3;        // 16-B chunks of 8 bf16
    const int total = halo_h * halo_w * chunks;
    for (int i = tid; i < total; i += 256) {
      const int chunk = i % chunks;
      const int pix = i / chunks;
      const int hrow = pix / halo_w, hcol = pix % halo_w;
      const int iy = oh0 - pad + hrow;
      const int ix = ow0 - pad + hcol;
      uint4 v = make_uint4(0, 0, 0, 0);
      if (iy >= 0 && iy < H && ix >= 0 && ix < W) {
        v = *reinterpret_cast<const uint4*>(
            x + (((long)img * H + iy) * W + ix) * C + chunk * 8);
      }
      *reinterpret_cast<uint4*>(
          &xtile[(hrow * HALO_W + hcol) * XPITCH + chunk * 8]) = v;
    }
  }

  // ---- prefetch weight chunk rs=0 ----
  // wpk layout bytes == LDS layout bytes (both [c16][n][WPAD]); glds
  // dest is wave-uniform base + lane*16, source is the matching linear
  // address.
  const int glds_per_chunk = (wchunk_bf16 * 2) / 16;  // 16-B pieces
  auto stage_w = [&](int rs, int buf) {
    const cbf16_t* src = wpk + (long)rs * wchunk_bf16;
    short* dst = wbuf(buf);
    for (int i = tid; i < glds_per_chunk; i += 256) {
      *reinterpret_cast<uint4*>(&dst[i * 8]) =
          *reinterpret_cast<const uint4*>(&src[i * 8]);
    }
  };
  stage_w(0, 0);
  __syncthreads();

  // ---- main loop over (r, s) ----
  cf32x16 acc[2] = {{}, {}};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
  // Wave's 32 pixels: rows [wave*2, wave*2+2) x 16 cols.
  const int prow = (wave * 32 + mrow) / TILE_W;
  const int pcol = (wave * 32 + mrow) % TILE_W;

  // T14-style register prefetch: issue chunk rs+1's global loads BEFORE
  // the rs MFMAs (latency hides under compute), write them to the spare
  // LDS buffer after the consumers' barrier.
  const int RS = R * S;
  uint4 wreg[3];                    // 3 x 16 B per thread covers 12 KiB
  const int wpieces = glds_per_chunk;  // <= 768
  auto fetch_w = [&](int rs) {
    const cbf16_t* src = wpk + (long)rs * wchunk_bf16;
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      const int i = tid + j * 256;
      if (i < wpieces)
        wreg[j] = *reinterpret_cast<const uint4*>(&src[i * 8]);
    }
  };
  auto write_w = [&](int buf) {
    short* dst = wbuf(buf);
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      const int i = tid + j * 256;
      if (i < wpieces)
        *reinterpret_cast<uint4*>(&dst[i * 8]) = wreg[j];
    }
  };
  for (int rs = 0; rs < RS; ++rs) {
    const int r = rs / S, s = rs % S;
    const int buf = rs & 1;
    if (rs + 1 < RS) fetch_w(rs + 1);  // loads in flight over the MFMAs
    for (int c16 = 0; c16 < c16n; ++c16) {
      cbf16x8 a_frag = *reinterpret_cast<const cbf16x8*>(
          &xtile[((prow + r) * HALO_W + (pcol + s)) * XPITCH
                 + c16 * 16 + kgrp * 8]);
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        if (nt >= ntiles) break;
        const int n = nt * 32 + mrow;
        cbf16x8 b_frag = *reinterpret_cast<const cbf16x8*>(
            &wbuf(buf)[(c16 * K + n) * WPAD + kgrp * 8]);
        acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_frag, b_frag, acc[nt], 0, 0, 0);
      }
    }
    if (rs + 1 < RS) {
      __syncthreads();              // wbuf[buf^1] consumers done
      write_w(buf ^ 1);
      __syncthreads();              // wbuf[buf^1] ready
    }
  }

  // ---- epilogue: scatter accumulators ----
  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
    if (nt >= ntiles) break;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / TILE_W;
      const int ocol = ow0 + p % TILE_W;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * K + nt * 32 + ocol_n] =
            __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Host wrapper
// ---------------------------------------------------------------------------

at::Tensor conv_s1_nhwc(at::Tensor x, at::Tensor wpk, int64_t K,
                        int64_t R, int64_t S, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_s1_nhwc: bf16 CUDA input required");
  TORCH_CHECK(x.dim() == 4 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_s1_nhwc: NCHW channels_last required");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 16 == 0 && C <= 64, "conv_s1_nhwc: C % 16, C <= 64");
  TORCH_CHECK(K % 32 == 0 && K <= 64, "conv_s1_nhwc: K % 32, K <= 64");
  TORCH_CHECK(R <= 5 && S <= 5, "conv_s1_nhwc: R,S <= 5");
  const int OH = H + 2 * pad - R + 1;
  const int OW = W + 2 * pad - S + 1;
  TORCH_CHECK(OH > 0 && OW > 0);
  auto y = at::empty({N, K, OH, OW},
                     x.options().memory_format(
                         at::MemoryFormat::ChannelsLast));
  const int tiles_h = (OH + TILE_H - 1) / TILE_H;
  const int tiles_w = (OW + TILE_W - 1) / TILE_W;
  const long grid = (long)N * tiles_h * tiles_w;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(conv_s1_nhwc_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     (const cbf16_t*)x.data_ptr(),
                     (const cbf16_t*)wpk.data_ptr(),
                     (cbf16_t*)y.data_ptr(),
                     N, C, H, W, (int)K, (int)R, (int)S, (int)pad,
                     OH, OW, tiles_h, tiles_w);
  return y;
}
