// Hand-written MFMA conv for stride-1 NHWC bf16 (gfx950 / CDNA4).
//
// Covers the Grasping44 stride-1 convs (5x5 SAME @79^2, 3x3 SAME/VALID,
// C=K=64) where MIOpen's igemm runs at ~15-20% MFMA utilization on
// these small-channel shapes (profiles/).  Backward-data is the same
// kernel on flipped/transposed prepacked weights.
//
// Design (cdna_hip_programming.md §3/§5, MI355X_MICROARCH.md):
//  * implicit GEMM: out[p, k] = sum_{r,s,c} x[p+D(r,s), c] * w[r,s,c,k],
//    v_mfma_f32_32x32x16_bf16; M = pixels, N = K, K' = C*R*S.
//  * a 256-thread WG computes an 8x16-pixel tile x K_WG channels:
//    4 waves x (32 pixels x K_WG).  The halo tile lives in LDS with a
//    PADDED 144-B pixel stride (linear 128-B stride = 16-way bank
//    conflict on the A-fragment ds_read_b128 -- the §5 GEMM trap).
//  * ALL weight chunks for the WG's channel group are staged ONCE
//    ([rs][c16][n][16] image, 32-B n-stride = 2-way conflict), so the
//    main loop is barrier-free.  When R*S*C*K_WG exceeds LDS the K dim
//    splits across blockIdx.y (5x5: K_WG=32, 2 splits, ~137 KiB LDS).
//  * measured (same-box interleaved A/B vs MIOpen/CK):
//    3x3@27^2 3.4x, 3x3@14^2 3.7x; 5x5@79^2 -- see profiles/.
//  * fragment layouts verified by the mfma_probe GPU test:
//    A row = l%32, k = (l>>5)*8+j; B col = l%32 (same k map);
//    C/D col = l&31, row = (reg&3)+8*(reg>>2)+4*(l>>5).
//  * all LDS in ONE __shared__ array (§5 trap 4a).

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
#define XPITCH 72             // bf16 per pixel row in LDS (64 + 8 pad)
#define WPAD 24               // bf16 per n-row in the GLOBAL packed image

#define XTILE_BF16 (HALO_H * HALO_W * XPITCH)

// C16N: C/16.  NT_WG: 32-wide N tiles computed per workgroup.
// RS_CAP: compile-time R*S capacity of the staged weight image.
template <int C16N, int NT_WG, int RS_CAP>
__global__ void __launch_bounds__(256, 2)
conv_s1_nhwc_kernel(const cbf16_t* __restrict__ x,
                    const cbf16_t* __restrict__ wpk,
                    cbf16_t* __restrict__ y,
                    int N, int H, int W, int K,
                    int R, int S, int pad,
                    int OH, int OW, int tiles_h, int tiles_w) {
  constexpr int C = C16N * 16;
  constexpr int K_WG = NT_WG * 32;
  constexpr int WLDS = RS_CAP * C16N * K_WG * 16;
  __shared__ short lds[XTILE_BF16 + WLDS];
  short* xtile = lds;
  short* wall = lds + XTILE_BF16;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int n0 = blockIdx.y * K_WG;    // channel-group offset

  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int th = trest / tiles_w;
  const int tw = trest % tiles_w;
  const int oh0 = th * TILE_H, ow0 = tw * TILE_W;

  // ---- stage the x halo tile (zero OOB), 16-B chunks ----
  {
    const int halo_h = TILE_H + R - 1, halo_w = TILE_W + S - 1;
    constexpr int chunks = C >> 3;
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

  // ---- stage every weight chunk for this channel group ----
  // Global image: [rs][c16][n(K)][WPAD]; LDS image:
  // [rs][c16][n(K_WG)][16].  Two 16-B pieces per n-row.
  {
    const int rows = R * S * C16N * K_WG;
    for (int i = tid; i < rows * 2; i += 256) {
      const int nrow = i >> 1, half = (i & 1) * 8;
      const int nn = nrow % K_WG;
      const int rc = nrow / K_WG;          // rs * C16N + c16
      *reinterpret_cast<uint4*>(&wall[nrow * 16 + half]) =
          *reinterpret_cast<const uint4*>(
              &wpk[((long)rc * K + n0 + nn) * WPAD + half]);
    }
  }
  __syncthreads();

  // ---- barrier-free main loop ----
  cf32x16 acc[NT_WG];
#pragma unroll
  for (int nt = 0; nt < NT_WG; ++nt) acc[nt] = (cf32x16){};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
  const int prow = (wave * 32 + mrow) / TILE_W;
  const int pcol = (wave * 32 + mrow) % TILE_W;

  const int RS = R * S;
  for (int rs = 0; rs < RS; ++rs) {
    const int r = rs / S, s = rs % S;
#pragma unroll
    for (int c16 = 0; c16 < C16N; ++c16) {
      cbf16x8 a_frag = *reinterpret_cast<const cbf16x8*>(
          &xtile[((prow + r) * HALO_W + (pcol + s)) * XPITCH
                 + c16 * 16 + kgrp * 8]);
#pragma unroll
      for (int nt = 0; nt < NT_WG; ++nt) {
        const int n = nt * 32 + mrow;
        cbf16x8 b_frag = *reinterpret_cast<const cbf16x8*>(
            &wall[(((rs * C16N) + c16) * K_WG + n) * 16 + kgrp * 8]);
        acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_frag, b_frag, acc[nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: scatter accumulators ----
  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < NT_WG; ++nt) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / TILE_W;
      const int ocol = ow0 + p % TILE_W;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * K
          + n0 + nt * 32 + ocol_n] = __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Big-tile variant for large-RS convs (5x5 @ 79^2): 512 threads compute a
// 16x16-pixel tile x all 64 channels; weights double-buffered per (r,s)
// with register prefetch (2x the compute per staged weight byte of the
// 256-thread tile; staging loads hide under the MFMAs).
// ---------------------------------------------------------------------------

#define BTILE 16
#define BHALO (BTILE + 4)

template <int C16N, int NTILES>
__global__ void __launch_bounds__(512, 2)
conv_s1_nhwc_big_kernel(const cbf16_t* __restrict__ x,
                        const cbf16_t* __restrict__ wpk,
                        cbf16_t* __restrict__ y,
                        int N, int H, int W, int K,
                        int R, int S, int pad,
                        int OH, int OW, int tiles_h, int tiles_w) {
  constexpr int C = C16N * 16;
  constexpr int WBUF = C16N * NTILES * 32 * WPAD;     // bf16 per chunk
  __shared__ short lds[BHALO * BHALO * XPITCH + 2 * WBUF];
  short* xtile = lds;
  auto wbuf = [&](int b) -> short* {
    return lds + BHALO * BHALO * XPITCH + b * WBUF;
  };

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int th = trest / tiles_w;
  const int tw = trest % tiles_w;
  const int oh0 = th * BTILE, ow0 = tw * BTILE;

  {
    const int halo_h = BTILE + R - 1, halo_w = BTILE + S - 1;
    constexpr int chunks = C >> 3;
    const int total = halo_h * halo_w * chunks;
    for (int i = tid; i < total; i += 512) {
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
          &xtile[(hrow * BHALO + hcol) * XPITCH + chunk * 8]) = v;
    }
  }

  constexpr int WPIECES = (WBUF * 2) / 16;
  constexpr int WPT = (WPIECES + 511) / 512;
  {
    const cbf16_t* src = wpk;
    short* dst = wbuf(0);
#pragma unroll
    for (int j = 0; j < WPT; ++j) {
      const int i = tid + j * 512;
      if (i < WPIECES)
        *reinterpret_cast<uint4*>(&dst[i * 8]) =
            *reinterpret_cast<const uint4*>(&src[i * 8]);
    }
  }
  __syncthreads();

  cf32x16 acc[NTILES];
#pragma unroll
  for (int nt = 0; nt < NTILES; ++nt) acc[nt] = (cf32x16){};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
  const int prow = (wave * 32 + mrow) / BTILE;
  const int pcol = (wave * 32 + mrow) % BTILE;

  const int RS = R * S;
  constexpr int wchunk_bf16 = WBUF;
  uint4 wreg[WPT];
  for (int rs = 0; rs < RS; ++rs) {
    const int r = rs / S, s = rs % S;
    const int buf = rs & 1;
    if (rs + 1 < RS) {
      const cbf16_t* src = wpk + (long)(rs + 1) * wchunk_bf16;
#pragma unroll
      for (int j = 0; j < WPT; ++j) {
        const int i = tid + j * 512;
        if (i < WPIECES)
          wreg[j] = *reinterpret_cast<const uint4*>(&src[i * 8]);
      }
    }
#pragma unroll
    for (int c16 = 0; c16 < C16N; ++c16) {
      cbf16x8 a_frag = *reinterpret_cast<const cbf16x8*>(
          &xtile[((prow + r) * BHALO + (pcol + s)) * XPITCH
                 + c16 * 16 + kgrp * 8]);
#pragma unroll
      for (int nt = 0; nt < NTILES; ++nt) {
        const int n = nt * 32 + mrow;
        cbf16x8 b_frag = *reinterpret_cast<const cbf16x8*>(
            &wbuf(buf)[(c16 * K + n) * WPAD + kgrp * 8]);
        acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_frag, b_frag, acc[nt], 0, 0, 0);
      }
    }
    if (rs + 1 < RS) {
      __syncthreads();
      {
        short* dst = wbuf(buf ^ 1);
#pragma unroll
        for (int j = 0; j < WPT; ++j) {
          const int i = tid + j * 512;
          if (i < WPIECES)
            *reinterpret_cast<uint4*>(&dst[i * 8]) = wreg[j];
        }
      }
      __syncthreads();
    }
  }

  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < NTILES; ++nt) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / BTILE;
      const int ocol = ow0 + p % BTILE;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * K
          + nt * 32 + ocol_n] = __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}

template <int C16N, int NTILES, int RING_DEPTH, int WAVES = 8,
          bool SETPRIO = false>
__global__ void conv_s1_nhwc_ring_kernel(
    const cbf16_t* __restrict__ x, const cbf16_t* __restrict__ wpk,
    cbf16_t* __restrict__ y, int N, int H, int W, int K, int R, int S,
    int pad, int OH, int OW, int tiles_h, int tiles_w);

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
  const bool small = (R * S) <= 9;
  auto stream = at::cuda::getCurrentCUDAStream();
  if (small) {
    const int tiles_h = (OH + TILE_H - 1) / TILE_H;
    const int tiles_w = (OW + TILE_W - 1) / TILE_W;
    const long grid = (long)N * tiles_h * tiles_w;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, stream.stream(),
                         (const cbf16_t*)x.data_ptr(),
                         (const cbf16_t*)wpk.data_ptr(),
                         (cbf16_t*)y.data_ptr(),
                         N, H, W, (int)K, (int)R, (int)S, (int)pad,
                         OH, OW, tiles_h, tiles_w);
    };
    if (C == 64 && K == 64) launch(conv_s1_nhwc_kernel<4, 2, 9>);
    else if (C == 32 && K == 32) launch(conv_s1_nhwc_kernel<2, 1, 9>);
    else if (C == 48 && K == 64) launch(conv_s1_nhwc_kernel<3, 2, 9>);
    else if (C == 16 && K == 64) launch(conv_s1_nhwc_kernel<1, 2, 9>);
    else if (C == 16 && K == 32) launch(conv_s1_nhwc_kernel<1, 1, 9>);
    else TORCH_CHECK(false, "conv_s1_nhwc: unsupported C/K combo");
  } else {
    const int tiles_h = (OH + BTILE - 1) / BTILE;
    const int tiles_w = (OW + BTILE - 1) / BTILE;
    const long grid = (long)N * tiles_h * tiles_w;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(grid), dim3(512), 0, stream.stream(),
                         (const cbf16_t*)x.data_ptr(),
                         (const cbf16_t*)wpk.data_ptr(),
                         (cbf16_t*)y.data_ptr(),
                         N, H, W, (int)K, (int)R, (int)S, (int)pad,
                         OH, OW, tiles_h, tiles_w);
    };
    static const bool use_ring = std::getenv("T2R_CONV_NO_RING") == nullptr;
    // T2R_RING_DEPTH: 3 (shallow, 94.5 KiB LDS; default — measured
    // 477 vs 446 TF against the 6-deep ring: the extra LDS slots buy
    // no latency cover at 1 WG/CU and cost L2 locality) or 6.
    static const bool deep_ring = []() {
      const char* v = std::getenv("T2R_RING_DEPTH");
      return v != nullptr && v[0] == '6';
    }();
    // 4-wave default (measured 490-497 vs 478 TF on the 5x5 @78^2):
    // 256-thread WGs on 8x16 tiles -> 2 independent WGs/CU whose
    // barriers don't couple, worth more than the 2x weight streaming.
    // T2R_RING_WAVES=8 selects the single 512-thread-WG variant.
    static const bool small_waves = []() {
      const char* v = std::getenv("T2R_RING_WAVES");
      return v == nullptr || v[0] != '8';
    }();
    if (C == 64 && K == 64 && use_ring && small_waves) {
      static const bool setprio = []() {
        const char* v = std::getenv("T2R_RING_SETPRIO");
        return v != nullptr && v[0] == '1';
      }();
      const int th8 = (OH + 7) / 8;
      const long grid8 = (long)N * th8 * tiles_w;
      if (setprio) {
        hipLaunchKernelGGL((conv_s1_nhwc_ring_kernel<4, 2, 3, 4, true>),
                           dim3(grid8), dim3(256), 0, stream.stream(),
                           (const cbf16_t*)x.data_ptr(),
                           (const cbf16_t*)wpk.data_ptr(),
                           (cbf16_t*)y.data_ptr(),
                           N, H, W, (int)K, (int)R, (int)S, (int)pad,
                           OH, OW, th8, tiles_w);
      } else {
        hipLaunchKernelGGL((conv_s1_nhwc_ring_kernel<4, 2, 3, 4, false>),
                           dim3(grid8), dim3(256), 0, stream.stream(),
                           (const cbf16_t*)x.data_ptr(),
                           (const cbf16_t*)wpk.data_ptr(),
                           (cbf16_t*)y.data_ptr(),
                           N, H, W, (int)K, (int)R, (int)S, (int)pad,
                           OH, OW, th8, tiles_w);
      }
    } else if (C == 64 && K == 64)
      use_ring ? (deep_ring ? launch(conv_s1_nhwc_ring_kernel<4, 2, 6>)
                            : launch(conv_s1_nhwc_ring_kernel<4, 2, 3>))
               : launch(conv_s1_nhwc_big_kernel<4, 2>);
    else if (C == 32 && K == 32)
      use_ring ? launch(conv_s1_nhwc_ring_kernel<2, 1, 6>)
               : launch(conv_s1_nhwc_big_kernel<2, 1>);
    else if (C == 48 && K == 64)
      use_ring ? launch(conv_s1_nhwc_ring_kernel<3, 2, 6>)
               : launch(conv_s1_nhwc_big_kernel<3, 2>);
    else if (C == 16 && K == 32)
      use_ring ? launch(conv_s1_nhwc_ring_kernel<1, 1, 6>)
               : launch(conv_s1_nhwc_big_kernel<1, 1>);
    else TORCH_CHECK(false, "conv_s1_nhwc: unsupported C/K combo");
  }
  return y;
}

// ---------------------------------------------------------------------------
// Single-kernel weight pack: [K,C,R,S] (torch, any float dtype widened to
// bf16 host-side = the .to(bf16) cast folds in here) -> the conv kernel's
// [rs][c16][n][WPAD] image.  transpose=true additionally swaps the C/K
// roles and flips r,s — the backward-data pack — so the python wrapper
// launches ONE kernel instead of ~7 tensor ops per conv per step.
// ---------------------------------------------------------------------------

template <bool TRANSPOSE>
__global__ void __launch_bounds__(256)
pack_conv_w_kernel(const cbf16_t* __restrict__ w,   // [K][C][R][S]
                   cbf16_t* __restrict__ out,       // [rs][c16][n][WPAD]
                   int K, int C, int R, int S) {
  // Output dims: C-role = TRANSPOSE ? K : C; K-role = TRANSPOSE ? C : K.
  const int crole = TRANSPOSE ? K : C;
  const int krole = TRANSPOSE ? C : K;
  const long total = (long)R * S * (crole / 16) * krole * WPAD;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int cc = i % WPAD;
    long rest = i / WPAD;
    const int n = rest % krole;
    rest /= krole;
    const int c16 = rest % (crole / 16);
    const int rs = rest / (crole / 16);
    cbf16_t v = __float2bfloat16(0.0f);
    if (cc < 16) {
      const int c = c16 * 16 + cc;
      int r = rs / S, s = rs % S;
      int kk, ci;
      if (TRANSPOSE) {
        kk = c;          // original K index comes from the C-role dim
        ci = n;          // original C index comes from the K-role dim
        r = R - 1 - r;
        s = S - 1 - s;
      } else {
        kk = n;
        ci = c;
      }
      v = w[(((long)kk * C + ci) * R + r) * S + s];
    }
    out[i] = v;
  }
}

at::Tensor pack_conv_w(at::Tensor w, bool transpose) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 4, "pack_conv_w: 4D CUDA");
  w = w.contiguous();
  if (w.scalar_type() != at::kBFloat16) w = w.to(at::kBFloat16);
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  const int crole = transpose ? K : C;
  const int krole = transpose ? C : K;
  TORCH_CHECK(crole % 16 == 0, "pack_conv_w: C-role % 16");
  auto out = at::empty({(long)R * S, crole / 16, krole, WPAD},
                       w.options());
  const long total = (long)R * S * (crole / 16) * krole * WPAD;
  const int grid = (int)std::min<long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (transpose)
    hipLaunchKernelGGL(pack_conv_w_kernel<true>, dim3(grid), dim3(256),
                       0, stream.stream(), (const cbf16_t*)w.data_ptr(),
                       (cbf16_t*)out.data_ptr(), K, C, R, S);
  else
    hipLaunchKernelGGL(pack_conv_w_kernel<false>, dim3(grid), dim3(256),
                       0, stream.stream(), (const cbf16_t*)w.data_ptr(),
                       (cbf16_t*)out.data_ptr(), K, C, R, S);
  return out;
}

__global__ void __launch_bounds__(256)
pack_conv_w_pair_kernel(const cbf16_t* __restrict__ w,  // [K][C][R][S]
                        cbf16_t* __restrict__ outf,
                        cbf16_t* __restrict__ outb,
                        int K, int C, int R, int S,
                        long totalf, long totalb) {
  for (long ii = blockIdx.x * (long)blockDim.x + threadIdx.x;
       ii < totalf + totalb; ii += (long)gridDim.x * blockDim.x) {
    const bool tr = ii >= totalf;
    const long i = tr ? ii - totalf : ii;
    const int crole = tr ? K : C;
    const int krole = tr ? C : K;
    const int cc = i % WPAD;
    long rest = i / WPAD;
    const int n = rest % krole;
    rest /= krole;
    const int c16 = rest % (crole / 16);
    const int rs = rest / (crole / 16);
    cbf16_t v = __float2bfloat16(0.0f);
    if (cc < 16) {
      const int c = c16 * 16 + cc;
      int r = rs / S, s = rs % S;
      int kk, ci;
      if (tr) { kk = c; ci = n; r = R - 1 - r; s = S - 1 - s; }
      else    { kk = n; ci = c; }
      v = w[(((long)kk * C + ci) * R + r) * S + s];
    }
    (tr ? outb : outf)[i] = v;
  }
}

std::vector<at::Tensor> pack_conv_w_pair(at::Tensor w) {
  // One dispatch emitting BOTH the forward pack and the dgrad pack
  // (transposed c/k roles, rotated rs) — halves the per-step pack
  // launches and takes the pack off the backward critical path.
  TORCH_CHECK(w.is_cuda() && w.dim() == 4, "pack_conv_w_pair: 4D CUDA");
  w = w.contiguous();
  if (w.scalar_type() != at::kBFloat16) w = w.to(at::kBFloat16);
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  TORCH_CHECK(C % 16 == 0 && K % 16 == 0, "pack_conv_w_pair: C,K % 16");
  auto outf = at::empty({(long)R * S, C / 16, K, WPAD}, w.options());
  auto outb = at::empty({(long)R * S, K / 16, C, WPAD}, w.options());
  const long totalf = (long)R * S * (C / 16) * K * WPAD;
  const long totalb = (long)R * S * (K / 16) * C * WPAD;
  const int grid =
      (int)std::min<long>((totalf + totalb + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pack_conv_w_pair_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(), (const cbf16_t*)w.data_ptr(),
                     (cbf16_t*)outf.data_ptr(),
                     (cbf16_t*)outb.data_ptr(), K, C, R, S,
                     totalf, totalb);
  return {outf, outb};
}

// ---------------------------------------------------------------------------
// glds-ring variant for the 5x5: weight chunks stream via LDS-DMA
// (global_load_lds) through a 3-slot ring with counted vmcnt + raw
// barriers (cdna_hip_programming.md §5 "glds, 2-3 LDS buffers" rows) —
// no register round-trip, no write pass, loads 2 chunks ahead.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void conv_waitcnt_vm(int count) {
  // s_waitcnt imm: vmcnt[3:0], expcnt[6:4]=7, lgkmcnt[11:8]=15.
  // `count` = allowed outstanding glds pieces for this wave.
  switch (count) {
    case 0: __builtin_amdgcn_s_waitcnt(0 | (7 << 4) | (15 << 8)); break;
    case 1: __builtin_amdgcn_s_waitcnt(1 | (7 << 4) | (15 << 8)); break;
    case 2: __builtin_amdgcn_s_waitcnt(2 | (7 << 4) | (15 << 8)); break;
    case 3: __builtin_amdgcn_s_waitcnt(3 | (7 << 4) | (15 << 8)); break;
    case 4: __builtin_amdgcn_s_waitcnt(4 | (7 << 4) | (15 << 8)); break;
    case 5: __builtin_amdgcn_s_waitcnt(5 | (7 << 4) | (15 << 8)); break;
    case 6: __builtin_amdgcn_s_waitcnt(6 | (7 << 4) | (15 << 8)); break;
    case 8: __builtin_amdgcn_s_waitcnt(8 | (7 << 4) | (15 << 8)); break;
    default:
      __builtin_amdgcn_s_waitcnt(10 | (7 << 4) | (15 << 8)); break;
  }
}

// WAVES=8: one 512-thread WG owns a 16x16 tile (1 WG/CU).  WAVES=4:
// a 256-thread WG owns an 8x16 tile -> 2 independent WGs/CU with
// uncoupled barriers, at the cost of streaming each weight chunk
// twice per 256 output pixels.
template <int C16N, int NTILES, int RING_DEPTH, int WAVES,
          bool SETPRIO>
__global__ void __launch_bounds__(WAVES * 64, 2)
conv_s1_nhwc_ring_kernel(const cbf16_t* __restrict__ x,
                         const cbf16_t* __restrict__ wpk,
                         cbf16_t* __restrict__ y,
                         int N, int H, int W, int K,
                         int R, int S, int pad,
                         int OH, int OW, int tiles_h, int tiles_w) {
  constexpr int C = C16N * 16;
  constexpr int WBUF = C16N * NTILES * 32 * WPAD;      // bf16 per chunk
  constexpr int PIECES = (WBUF * 2) / 1024;            // 1-KiB DMA pieces
  constexpr int NTHREADS = WAVES * 64;
  constexpr int TILE_HH = WAVES * 2;                   // tile = TH x 16
  constexpr int HALO_HH = TILE_HH + 4;
  __shared__ short lds[HALO_HH * BHALO * XPITCH + RING_DEPTH * WBUF];
  short* xtile = lds;
  auto wbuf = [&](int slot) -> short* {
    return lds + HALO_HH * BHALO * XPITCH + slot * WBUF;
  };

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  long wg = blockIdx.x;
  const int img = wg / (tiles_h * tiles_w);
  const int trest = wg % (tiles_h * tiles_w);
  const int th = trest / tiles_w;
  const int tw = trest % tiles_w;
  const int oh0 = th * TILE_HH, ow0 = tw * BTILE;

  // Uniform piece assignment so the counted vmcnt is the same for all
  // issuer waves: WAVES=8 -> 6 issuer waves x 2 pieces (PIECES=12);
  // WAVES=4 -> 4 issuer waves x PIECES/4.
  constexpr int ISSUERS = (WAVES == 8) ? 6 : WAVES;
  constexpr int PPW = (PIECES + ISSUERS - 1) / ISSUERS;
  const bool issuer = wave < ISSUERS;
  auto issue_chunk = [&](int rs, int slot) {
    if (!issuer) return;
#pragma unroll
    for (int j = 0; j < PPW; ++j) {
      const int piece = wave * PPW + j;
      if (piece < PIECES) {
        const cbf16_t* src = wpk + (long)rs * WBUF + piece * 512
                             + lane * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)src,
            (__attribute__((address_space(3))) uint32_t*)
                (wbuf(slot) + piece * 512),
            16, 0, 0);
      }
    }
  };

  {
    const int halo_h = TILE_HH + R - 1, halo_w = BTILE + S - 1;
    constexpr int chunks = C >> 3;
    const int total = halo_h * halo_w * chunks;
    for (int i = tid; i < total; i += NTHREADS) {
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
          &xtile[(hrow * BHALO + hcol) * XPITCH + chunk * 8]) = v;
    }
  }
  for (int cpre = 0; cpre < RING_DEPTH - 1 && cpre < R * S; ++cpre) {
    issue_chunk(cpre, cpre % RING_DEPTH);
  }
  // One full drain in the prologue (also covers the x-tile loads).
  __syncthreads();

  cf32x16 acc[NTILES];
#pragma unroll
  for (int nt = 0; nt < NTILES; ++nt) acc[nt] = (cf32x16){};
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;
  const int prow = (wave * 32 + mrow) / BTILE;
  const int pcol = (wave * 32 + mrow) % BTILE;

  const int RS = R * S;
  for (int rs = 0; rs < RS; ++rs) {
    const int r = rs / S, s = rs % S;
    const int slot = rs % RING_DEPTH;
    if (rs + RING_DEPTH - 1 < RS)
      issue_chunk(rs + RING_DEPTH - 1, (rs + RING_DEPTH - 1) % RING_DEPTH);
    if (issuer)
      conv_waitcnt_vm(min(RS - 1 - rs, RING_DEPTH - 1) * PPW);
    __builtin_amdgcn_s_barrier();     // chunk rs landed for everyone
    // T5 (guide): favor the MFMA cluster while issuer waves run their
    // global_load_lds issue phase (A/B via T2R_RING_SETPRIO).
    if constexpr (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c16 = 0; c16 < C16N; ++c16) {
      cbf16x8 a_frag = *reinterpret_cast<const cbf16x8*>(
          &xtile[((prow + r) * BHALO + (pcol + s)) * XPITCH
                 + c16 * 16 + kgrp * 8]);
#pragma unroll
      for (int nt = 0; nt < NTILES; ++nt) {
        const int n = nt * 32 + mrow;
        cbf16x8 b_frag = *reinterpret_cast<const cbf16x8*>(
            &wbuf(slot)[(c16 * K + n) * WPAD + kgrp * 8]);
        acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_frag, b_frag, acc[nt], 0, 0, 0);
      }
    }
    if constexpr (SETPRIO) __builtin_amdgcn_s_setprio(0);
    // Slot rs%RING_DEPTH is refilled by the DMA issued at iteration
    // rs+1 (chunk rs+RING_DEPTH): everyone must be done reading
    // before that DMA can be issued.
    __builtin_amdgcn_s_barrier();
  }

  const int ocol_n = lane & 31;
#pragma unroll
  for (int nt = 0; nt < NTILES; ++nt) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int p = wave * 32 + m;
      const int orow = oh0 + p / BTILE;
      const int ocol = ow0 + p % BTILE;
      if (orow < OH && ocol < OW) {
        y[(((long)img * OH + orow) * OW + ocol) * K
          + nt * 32 + ocol_n] = __float2bfloat16(acc[nt][reg]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Fused space-to-depth for the 6x6/2 stem: [N][H][W][3] bf16 (NHWC) ->
// [N][H/2][W/2][16] with c' = dr*6 + ds*3 + c and slots 12..15 zero.
// One global read + one write replaces the torch zeros/permute/copy
// chain (4 kernels, ~3x the bytes).  The 6 taps of each (dr) row are
// contiguous in NHWC (x[n, 2oy+dr, 2ox..2ox+1, 0..2]) and every
// offset is u32-aligned (element offsets are multiples of 6), so the
// read is 2x3 dwords and the write is 2 dwordx4 per output pixel.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) unsigned int cuint4v;

__global__ void __launch_bounds__(256)
s2d_stem_kernel(const unsigned int* __restrict__ x,
                cuint4v* __restrict__ y,
                long total, int H, int W, int OH, int OW) {
  const int row_u32 = (W * 3) >> 1;   // u32 per input row (W even)
  for (long p = blockIdx.x * 256L + threadIdx.x; p < total;
       p += (long)gridDim.x * 256) {
    const int ox = (int)(p % OW);
    long rest = p / OW;
    const int oy = (int)(rest % OH);
    const long n = rest / OH;
    const long base = (((n * H + 2L * oy) * W) * 3 >> 1) + 3L * ox;
    const unsigned int a0 = x[base], a1 = x[base + 1], a2 = x[base + 2];
    const long b = base + row_u32;
    const unsigned int b0 = x[b], b1 = x[b + 1], b2 = x[b + 2];
    y[p * 2] = cuint4v{a0, a1, a2, b0};
    y[p * 2 + 1] = cuint4v{b1, b2, 0u, 0u};
  }
}

at::Tensor s2d_stem(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "s2d_stem: bf16 CUDA input required");
  TORCH_CHECK(x.dim() == 4 && x.size(1) == 3 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "s2d_stem: [N,3,H,W] channels_last required");
  const int N = x.size(0), H = x.size(2), W = x.size(3);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "s2d_stem: even H, W required");
  const int OH = H / 2, OW = W / 2;
  auto y = at::empty({N, 16, OH, OW},
                     x.options().memory_format(
                         at::MemoryFormat::ChannelsLast));
  const long total = (long)N * OH * OW;
  const long blocks = std::min((total + 255) / 256, 8192L);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(s2d_stem_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(),
                     (const unsigned int*)x.data_ptr(),
                     (cuint4v*)y.data_ptr(), total, H, W, OH, OW);
  return y;
}
