// MFMA weight-gradient (wrw) kernel, v2: register accumulation.
//
// Same GEMM view as v1 (conv_wrw.hip): dW[rs][c][k] = sum_p
// x[p + D(rs), c] * dy[p, k] with the pixel axis as contraction; both
// operands staged TRANSPOSED ([c][p] / [k][p]) in LDS per spatial
// window.  v1 measured 0.25-0.44x MIOpen because each (window, rs)
// iteration did an LDS read-modify-write of the dW slice (48 LDS ops
// per rs per wave) and the RS_GROUP grid dimension re-staged every
// window up to 5x.  v2 removes both:
//   - 8 waves per WG = (c-tile x k-tile quadrant) x (rs half); each
//     wave keeps ITS rs-half's accumulators entirely in VGPRs
//     (ceil(25/2)=13 x 16 f32 = 208 VGPRs for the 5x5 -> 2 waves/SIMD)
//   - every window is staged exactly once; all rs offsets consume the
//     same LDS image (the halo covers every shift)
//   - one atomicAdd flush per WG at the end, straight from VGPRs.
//
// LDS: xT 64 x 248 x 2B = 31.7 KiB + dyT 64 x 136 x 2B = 17.4 KiB =
// 49.1 KiB -> one 512-thread WG per CU (VGPR-bound anyway).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 w2bf16_t;
typedef __attribute__((ext_vector_type(8))) short w2bf16x8;
typedef __attribute__((ext_vector_type(16))) float w2f32x16;

#define W2TILE_W 16
#define W2HALO_W (W2TILE_W + 4)

// TH = window tile height: 8 (128-pixel window, 49 KiB LDS) for the
// 3x3; 16 (256-pixel window, 86 KiB LDS) for the 5x5 — the bigger
// window doubles MFMAs per staged byte, which the 25-offset reuse
// needs to go compute-bound.
template <int R, int S, int TH>
__global__ void __launch_bounds__(512, 1)
conv_s1_wrw2_kernel(const w2bf16_t* __restrict__ x,
                    const w2bf16_t* __restrict__ dy,
                    float* __restrict__ dw,   // [R*S][C][K] f32, zeroed
                    int N, int H, int W, int pad,
                    int OH, int OW, int tiles_h, int tiles_w,
                    int window_groups) {
  constexpr int C = 64, K = 64;
  constexpr int RS = R * S;
  constexpr int RS_PER = (RS + 1) / 2;     // rs offsets per wave half
  constexpr int W2TILE_H = TH;
  constexpr int W2WIN_P = W2TILE_H * W2TILE_W;
  constexpr int W2HALO_H = W2TILE_H + 4;
  constexpr int W2HALO_P = W2HALO_H * W2HALO_W;
  constexpr int W2XT_PITCH = W2HALO_P + 8;     // multiple of 8 (b128)
  constexpr int W2DYT_PITCH = W2WIN_P + 8;
  __shared__ short lds_xt[C * W2XT_PITCH];
  __shared__ short lds_dyt[K * W2DYT_PITCH];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int mtile = wave & 1;              // c-tile
  const int ntile = (wave >> 1) & 1;       // k-tile
  const int rshalf = wave >> 2;            // 0 or 1
  const int rs_base = rshalf * RS_PER;

  w2f32x16 acc[RS_PER];
#pragma unroll
  for (int g = 0; g < RS_PER; ++g) acc[g] = w2f32x16{};

  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;

  for (int win = blockIdx.x; win < total_windows; win += window_groups) {
    const int img = win / (tiles_h * tiles_w);
    const int trest = win % (tiles_h * tiles_w);
    const int oh0 = (trest / tiles_w) * W2TILE_H;
    const int ow0 = (trest % tiles_w) * W2TILE_W;

    // ---- stage x halo transposed: lds_xt[c][halo_p] ----
    {
      constexpr int halo_h = W2TILE_H + R - 1, halo_w = W2TILE_W + S - 1;
      constexpr int chunks = C >> 3;
      for (int i = tid; i < W2HALO_P * chunks; i += 512) {
        const int chunk = i % chunks;
        const int p = i / chunks;
        const int hrow = p / W2HALO_W, hcol = p % W2HALO_W;
        const int iy = oh0 - pad + hrow;
        const int ix = ow0 - pad + hcol;
        w2bf16x8 v = {};
        if (hrow < halo_h && hcol < halo_w &&
            iy >= 0 && iy < H && ix >= 0 && ix < W) {
          v = *reinterpret_cast<const w2bf16x8*>(
              x + (((long)img * H + iy) * W + ix) * C + chunk * 8);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_xt[(chunk * 8 + j) * W2XT_PITCH + p] = v[j];
      }
    }
    // ---- stage dy window transposed: lds_dyt[k][win_p] ----
    {
      constexpr int kchunks = K >> 3;
      for (int i = tid; i < W2WIN_P * kchunks; i += 512) {
        const int chunk = i % kchunks;
        const int p = i / kchunks;
        const int orow = oh0 + p / W2TILE_W;
        const int ocol = ow0 + p % W2TILE_W;
        w2bf16x8 v = {};
        if (orow < OH && ocol < OW) {
          v = *reinterpret_cast<const w2bf16x8*>(
              dy + (((long)img * OH + orow) * OW + ocol) * K + chunk * 8);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_dyt[(chunk * 8 + j) * W2DYT_PITCH + p] = v[j];
      }
    }
    __syncthreads();

    // ---- all rs offsets of this wave's half, window staged once.
    // kstep outer so the dy fragment (independent of rs) is read once
    // and reused across the whole rs half: 8 b-reads + RS_PER*8
    // a-reads per wave per window instead of 2*RS_PER*8. ----
    {
      const int c = mtile * 32 + mrow;
      const int k = ntile * 32 + mrow;
      if constexpr (RS_PER <= 5) {
        // kstep outer: the dy fragment (independent of rs) is read
        // once per kstep and reused across the rs half (8 b-reads +
        // RS_PER*8 a-reads instead of 2*RS_PER*8).  Register-safe at
        // RS_PER<=5 (149 VGPRs, no spills).
#pragma unroll
        for (int kstep = 0; kstep < W2WIN_P / 16; ++kstep) {
          w2bf16x8 b_frag = *reinterpret_cast<const w2bf16x8*>(
              &lds_dyt[k * W2DYT_PITCH + kstep * 16 + kgrp * 8]);
#pragma unroll
          for (int g = 0; g < RS_PER; ++g) {
            const int rs = rs_base + g;
            if (rs < RS) {
              const int r = rs / S, s = rs % S;
              const int halo_base = (kstep + r) * W2HALO_W + s;
              w2bf16x8 a_frag = *reinterpret_cast<const w2bf16x8*>(
                  &lds_xt[c * W2XT_PITCH + halo_base + kgrp * 8]);
              acc[g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  a_frag, b_frag, acc[g], 0, 0, 0);
            }
          }
        }
      } else {
        // g outer: at RS_PER=13 the hoisted-b order spills 82 VGPRs
        // (13 concurrent a-load chains); one-g-at-a-time keeps the
        // 208 accumulators + one chain live (1 spill).
#pragma unroll
        for (int g = 0; g < RS_PER; ++g) {
          const int rs = rs_base + g;
          if (rs < RS) {
            const int r = rs / S, s = rs % S;
#pragma unroll
            for (int kstep = 0; kstep < W2WIN_P / 16; ++kstep) {
              const int halo_base = (kstep + r) * W2HALO_W + s;
              w2bf16x8 a_frag = *reinterpret_cast<const w2bf16x8*>(
                  &lds_xt[c * W2XT_PITCH + halo_base + kgrp * 8]);
              w2bf16x8 b_frag = *reinterpret_cast<const w2bf16x8*>(
                  &lds_dyt[k * W2DYT_PITCH + kstep * 16 + kgrp * 8]);
              acc[g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  a_frag, b_frag, acc[g], 0, 0, 0);
            }
          }
        }
      }
    }
    __syncthreads();   // all waves done before restaging
  }

  // ---- flush: non-atomic per-WG partial (atomics measured ~120 us
  // chip-wide: every WG RMWs the same 3200 cache lines; partials +
  // a reduce kernel are plain streaming traffic instead) ----
  float* part = dw + (long)blockIdx.x * RS * C * K;
#pragma unroll
  for (int g = 0; g < RS_PER; ++g) {
    const int rs = rs_base + g;
    if (rs < RS) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int c = mtile * 32
            + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int k = ntile * 32 + (lane & 31);
        part[((long)rs * C + c) * K + k] = acc[g][reg];
      }
    }
  }
}

__global__ void __launch_bounds__(256)
wrw2_reduce_kernel(const float* __restrict__ part,
                   float* __restrict__ dw, long cells, int nparts) {
  for (long i = blockIdx.x * 256L + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    float s = 0.0f;
    for (int p = 0; p < nparts; ++p) s += part[p * cells + i];
    dw[i] = s;
  }
}


// ---------------------------------------------------------------------------
// v3: occupancy-first split.  The v2 monolith (8 waves, 208 accum
// VGPRs, 2 waves/SIMD) is latency-bound; v3 follows the 4-wave ring
// lesson: 256-thread WGs where each wave is one (c,k) quadrant and
// blockIdx.y picks an rs GROUP of <=5 offsets -> 80 accumulator VGPRs
// and 3 WGs/CU (LDS-bound), so 3 waves/SIMD hide the MFMA + LDS
// latency.  Price: each window is staged once PER rs group (up to 5x
// global re-read, ~47 us chip-wide for the 5x5 @78^2 layer - cheap
// next to the latency win).  Flush: non-atomic per-WG partial slices.
// ---------------------------------------------------------------------------

#define W3_GROUP 5

template <int R, int S>
__global__ void __launch_bounds__(256, 2)
conv_s1_wrw3_kernel(const w2bf16_t* __restrict__ x,
                    const w2bf16_t* __restrict__ dy,
                    float* __restrict__ dw_part,  // [gx][gy][G][C][K]
                    int N, int H, int W, int pad,
                    int OH, int OW, int tiles_h, int tiles_w,
                    int window_groups) {
  constexpr int C = 64, K = 64;
  constexpr int RS = R * S;
  constexpr int TH = 8;
  constexpr int WIN_P = TH * W2TILE_W;         // 128
  constexpr int HALO_H = TH + 4;
  constexpr int HALO_P = HALO_H * W2HALO_W;    // 240
  constexpr int XT_P = HALO_P + 8;
  constexpr int DYT_P = WIN_P + 8;
  __shared__ short lds_xt[C * XT_P];
  __shared__ short lds_dyt[K * DYT_P];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int mtile = wave & 1;
  const int ntile = wave >> 1;
  const int rs_base = blockIdx.y * W3_GROUP;
  const int rs_count = min(RS - rs_base, W3_GROUP);

  w2f32x16 acc[W3_GROUP];
#pragma unroll
  for (int g = 0; g < W3_GROUP; ++g) acc[g] = w2f32x16{};

  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;

  for (int win = blockIdx.x; win < total_windows; win += window_groups) {
    const int img = win / (tiles_h * tiles_w);
    const int trest = win % (tiles_h * tiles_w);
    const int oh0 = (trest / tiles_w) * TH;
    const int ow0 = (trest % tiles_w) * W2TILE_W;

    // Two-phase staging: ALL guarded global loads are issued first
    // (independent, overlapped latencies), then the transposed LDS
    // write pass drains them with one wait — the single-loop
    // load->8-scattered-writes form serialized a full HBM/L2 latency
    // per iteration under this kernel's register pressure.
    {
      constexpr int halo_h = TH + R - 1, halo_w = W2TILE_W + S - 1;
      constexpr int chunks = C >> 3;
      constexpr int XITER = (HALO_P * chunks + 255) / 256;
      w2bf16x8 vx[XITER];
#pragma unroll
      for (int t = 0; t < XITER; ++t) {
        const int i = tid + t * 256;
        const int chunk = i % chunks;
        const int p = i / chunks;
        const int hrow = p / W2HALO_W, hcol = p % W2HALO_W;
        const int iy = oh0 - pad + hrow;
        const int ix = ow0 - pad + hcol;
        vx[t] = w2bf16x8{};
        if (i < HALO_P * chunks && hrow < halo_h && hcol < halo_w &&
            iy >= 0 && iy < H && ix >= 0 && ix < W) {
          vx[t] = *reinterpret_cast<const w2bf16x8*>(
              x + (((long)img * H + iy) * W + ix) * C + chunk * 8);
        }
      }
#pragma unroll
      for (int t = 0; t < XITER; ++t) {
        const int i = tid + t * 256;
        if (i < HALO_P * chunks) {
          const int chunk = i % chunks;
          const int p = i / chunks;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            lds_xt[(chunk * 8 + j) * XT_P + p] = vx[t][j];
        }
      }
    }
    {
      constexpr int kchunks = K >> 3;
      constexpr int YITER = (WIN_P * kchunks + 255) / 256;
      w2bf16x8 vy[YITER];
#pragma unroll
      for (int t = 0; t < YITER; ++t) {
        const int i = tid + t * 256;
        const int chunk = i % kchunks;
        const int p = i / kchunks;
        const int orow = oh0 + p / W2TILE_W;
        const int ocol = ow0 + p % W2TILE_W;
        vy[t] = w2bf16x8{};
        if (i < WIN_P * kchunks && orow < OH && ocol < OW) {
          vy[t] = *reinterpret_cast<const w2bf16x8*>(
              dy + (((long)img * OH + orow) * OW + ocol) * K + chunk * 8);
        }
      }
#pragma unroll
      for (int t = 0; t < YITER; ++t) {
        const int i = tid + t * 256;
        if (i < WIN_P * kchunks) {
          const int chunk = i % kchunks;
          const int p = i / kchunks;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            lds_dyt[(chunk * 8 + j) * DYT_P + p] = vy[t][j];
        }
      }
    }
    __syncthreads();

    {
      const int c = mtile * 32 + mrow;
      const int k = ntile * 32 + mrow;
#pragma unroll
      for (int kstep = 0; kstep < WIN_P / 16; ++kstep) {
        w2bf16x8 b_frag = *reinterpret_cast<const w2bf16x8*>(
            &lds_dyt[k * DYT_P + kstep * 16 + kgrp * 8]);
#pragma unroll
        for (int g = 0; g < W3_GROUP; ++g) {
          const int rs = rs_base + g;
          if (rs < RS) {
            const int r = rs / S, s = rs % S;
            const int halo_base = (kstep + r) * W2HALO_W + s;
            // Element-wise u16 reads: the (r,s)-shifted start is NOT
            // 16B aligned for most offsets, and a misaligned
            // ds_read_b128 replays at 64 cycles/instr (guide G17) —
            // the dominant stall of wrw v1-v3.
            const short* arow = &lds_xt[c * XT_P + halo_base + kgrp * 8];
            w2bf16x8 a_frag;
#pragma unroll
            for (int j = 0; j < 8; ++j) a_frag[j] = arow[j];
            acc[g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a_frag, b_frag, acc[g], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  float* part = dw_part
      + ((long)blockIdx.x * gridDim.y + blockIdx.y) * W3_GROUP * C * K;
#pragma unroll
  for (int g = 0; g < W3_GROUP; ++g) {
    if (rs_base + g < RS) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int c = mtile * 32
            + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int k = ntile * 32 + (lane & 31);
        part[((long)g * C + c) * K + k] = acc[g][reg];
      }
    }
  }
}

// Reduce: dw[rs][c][k] = sum over gx of part[gx][rs/G][rs%G][c][k].
__global__ void __launch_bounds__(256)
wrw3_reduce_kernel(const float* __restrict__ part,
                   float* __restrict__ dw, int RS, long ck,
                   int ngx, int ngy) {
  const long cells = (long)RS * ck;
  for (long i = blockIdx.x * 256L + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int rs = (int)(i / ck);
    const long rest = i % ck;
    const int gy = rs / W3_GROUP, g = rs % W3_GROUP;
    float s = 0.0f;
    for (int gx = 0; gx < ngx; ++gx) {
      s += part[(((long)gx * ngy + gy) * W3_GROUP + g) * ck + rest];
    }
    dw[i] = s;
  }
}

at::Tensor conv_s1_wrw3(at::Tensor x, at::Tensor dy, int64_t R,
                        int64_t S, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_s1_wrw3: bf16 x required");
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "conv_s1_wrw3: bf16 dy required");
  x = x.contiguous(at::MemoryFormat::ChannelsLast);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(C == 64 && K == 64, "conv_s1_wrw3: C == K == 64 only");
  TORCH_CHECK(OH == H + 2 * pad - R + 1 && OW == W + 2 * pad - S + 1,
              "conv_s1_wrw3: dy shape mismatch");
  const int tiles_h = (OH + 7) / 8;
  const int tiles_w = (OW + W2TILE_W - 1) / W2TILE_W;
  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int ngy = (int)((R * S + W3_GROUP - 1) / W3_GROUP);
  // 3 WGs/CU resident; split windows so the whole grid fills the chip
  // ~3x over without inflating the reduce.
  const int window_groups = std::min(total_windows, 768 / ngy * 1);
  auto part = at::empty({(long)window_groups * ngy, W3_GROUP,
                         (long)C, K}, x.options().dtype(at::kFloat));
  auto dw = at::empty({(long)R * S, C, K},
                      x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(window_groups, ngy), dim3(256),
                       0, stream.stream(),
                       (const w2bf16_t*)x.data_ptr(),
                       (const w2bf16_t*)dy.data_ptr(),
                       (float*)part.data_ptr(),
                       N, H, W, (int)pad, OH, OW, tiles_h, tiles_w,
                       window_groups);
  };
  if (R == 3 && S == 3) launch(conv_s1_wrw3_kernel<3, 3>);
  else if (R == 5 && S == 5) launch(conv_s1_wrw3_kernel<5, 5>);
  else TORCH_CHECK(false, "conv_s1_wrw3: R/S must be 3x3 or 5x5");
  const long cells = (long)R * S * C * K;
  const int rblocks = (int)std::min((cells + 255) / 256, 1024L);
  hipLaunchKernelGGL(wrw3_reduce_kernel, dim3(rblocks), dim3(256), 0,
                     stream.stream(), (const float*)part.data_ptr(),
                     (float*)dw.data_ptr(), (int)(R * S), (long)C * K,
                     window_groups, ngy);
  return dw;
}

at::Tensor conv_s1_wrw2(at::Tensor x, at::Tensor dy, int64_t R,
                        int64_t S, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_s1_wrw2: bf16 x required");
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "conv_s1_wrw2: bf16 dy required");
  x = x.contiguous(at::MemoryFormat::ChannelsLast);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(C == 64 && K == 64, "conv_s1_wrw2: C == K == 64 only");
  TORCH_CHECK(OH == H + 2 * pad - R + 1 && OW == W + 2 * pad - S + 1,
              "conv_s1_wrw2: dy shape mismatch");
  const long cells = (long)R * S * C * K;
  const int tile_h = (R == 5) ? 16 : 8;
  const int tiles_h = (OH + tile_h - 1) / tile_h;
  const int tiles_w = (OW + W2TILE_W - 1) / W2TILE_W;
  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  // One WG per CU (the kernel is VGPR-bound at 1 WG/CU anyway); each
  // WG owns a disjoint partial slice, summed by the reduce kernel.
  const int window_groups = std::min(total_windows, 256);
  auto part = at::empty({window_groups, (long)R * S, C, K},
                        x.options().dtype(at::kFloat));
  auto dw = at::empty({(long)R * S, C, K},
                      x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(window_groups), dim3(512), 0,
                       stream.stream(),
                       (const w2bf16_t*)x.data_ptr(),
                       (const w2bf16_t*)dy.data_ptr(),
                       (float*)part.data_ptr(),
                       N, H, W, (int)pad, OH, OW, tiles_h, tiles_w,
                       window_groups);
  };
  if (R == 3 && S == 3) launch(conv_s1_wrw2_kernel<3, 3, 8>);
  else if (R == 5 && S == 5) launch(conv_s1_wrw2_kernel<5, 5, 16>);
  else TORCH_CHECK(false, "conv_s1_wrw2: R/S must be 3x3 or 5x5");
  const int rblocks = (int)std::min((cells + 255) / 256, 1024L);
  hipLaunchKernelGGL(wrw2_reduce_kernel, dim3(rblocks), dim3(256), 0,
                     stream.stream(), (const float*)part.data_ptr(),
                     (float*)dw.data_ptr(), cells, window_groups);
  return dw;
}
