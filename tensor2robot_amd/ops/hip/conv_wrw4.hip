// MFMA weight-gradient (wrw) kernel, v4: pixel-major x image consumed
// with ds_read_b64_tr_b16 hardware transpose reads.
//
// Same GEMM view as v1-v3 (conv_wrw.hip / conv_wrw2.hip): dW[rs][c][k]
// = sum_p x[p + D(rs), c] * dy[p, k], pixel axis as contraction.
// v1-v3 all measured 0.25-0.5x MIOpen with the stall fingered at the
// TRANSPOSED x image: the (r,s)-shifted A-fragment start is not 16-B
// aligned for most offsets, so v3 fell back to element-wise u16 LDS
// reads (8 instructions per fragment, conflicted).  v4 removes the
// transpose entirely:
//
//  - x halo is staged PIXEL-major in packed [ph/4][c/16][4][16]
//    subtiles (one 16-B vector write per 8-channel chunk, no scatter).
//  - Each A fragment is TWO ds_read_b64_tr_b16: per 16-lane group the
//    instruction reads a [4 pixel][16 channel] row-major block and
//    delivers lane (l&15) its channel-column of 4 pixels; pixels +4..7
//    come from the next subtile block, exactly +512 B, so the second
//    read is the same address with offset:512.  Every lane address is
//    8-B aligned by construction (the G17 tr_b16 trap: a base = 2/4/6
//    mod 8 shorts silently returns the 8-aligned address's data).
//  - The block layout makes the 16 lane addresses of a group cover 32
//    distinct banks for EVERY (r,s) shift (pixel row = 32 B, block =
//    128 B: bank = 32*(sc&1) + (ph&3)*8 + piece*2, all distinct), i.e.
//    one of the guide's conflict-free tr subtilings.
//
// dy stays in the v3 transposed image read with aligned b128 (its
// fragments are always 16-B aligned).  Grid/flush structure follows v3
// (256-thread WGs = 4 (c,k)-quadrant waves, blockIdx.y = rs group of
// <=5 offsets, non-atomic per-WG partials + reduce kernel).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 w4bf16_t;
typedef __attribute__((ext_vector_type(8))) short w4bf16x8;
typedef __attribute__((ext_vector_type(4))) short w4short4;
typedef __attribute__((ext_vector_type(16))) float w4f32x16;

#define W4_TILE_W 16
#define W4_GROUP 5

template <int R, int S>
__global__ void __launch_bounds__(256, 3)
conv_s1_wrw4_kernel(const w4bf16_t* __restrict__ x,
                    const w4bf16_t* __restrict__ dy,
                    float* __restrict__ dw_part,  // [gx][gy][G][C][K]
                    int N, int H, int W, int pad,
                    int OH, int OW, int tiles_h, int tiles_w,
                    int window_groups) {
  constexpr int C = 64, K = 64;
  constexpr int RS = R * S;
  constexpr int TH = 8;
  constexpr int WIN_P = TH * W4_TILE_W;          // 128
  constexpr int HALO_H = TH + R - 1;             // 10 / 12
  constexpr int HALO_W = W4_TILE_W + S - 1;      // 18 / 20
  constexpr int HALO_P = HALO_H * HALO_W;        // 180 / 240 (%4 == 0)
  static_assert(HALO_P % 4 == 0, "halo pixel count must be 4-aligned");
  constexpr int DYT_P = WIN_P + 8;
  // x image: element (ph, c) lives at
  //   ((ph>>2)*4 + (c>>4))*64 + (ph&3)*16 + (c&15).
  // 16-B aligned base: tr_b16 lane addresses derive from it and a base
  // = 2/4/6 mod 8 shorts silently reads wrong data (guide G17).
  __shared__ __attribute__((aligned(16))) short lds_x[HALO_P * C];
  __shared__ __attribute__((aligned(16))) short lds_dyt[K * DYT_P];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int mtile = wave & 1;              // c half
  const int ntile = wave >> 1;             // k half
  const int rs_base = blockIdx.y * W4_GROUP;

  w4f32x16 acc[W4_GROUP];
#pragma unroll
  for (int g = 0; g < W4_GROUP; ++g) acc[g] = w4f32x16{};

  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;

  // tr_b16 lane address pieces (constant over the window loop): lane
  // l of 16-lane group grp covers pixel ph_base + (grp>>1)*8 +
  // ((l&15)>>2) at channel piece mtile*32 + (grp&1)*16 + ((l&15)&3)*4.
  const int grp = lane >> 4;
  const int m16 = lane & 15;
  const int pix_lane = (grp >> 1) * 8 + (m16 >> 2);   // pixel offset
  const int cpiece = mtile * 32 + (grp & 1) * 16 + (m16 & 3) * 4;

  for (int win = blockIdx.x; win < total_windows; win += window_groups) {
    const int img = win / (tiles_h * tiles_w);
    const int trest = win % (tiles_h * tiles_w);
    const int oh0 = (trest / tiles_w) * TH;
    const int ow0 = (trest % tiles_w) * W4_TILE_W;

    // ---- stage x halo pixel-major (two-phase: all guarded loads
    // issued first, then the write pass drains them).  Lane->(ph,
    // chunk) mapping staggers (ph&3, chunk&1) across each 8-lane
    // write-service group so the b128 writes are bank-conflict-free.
    {
      constexpr int ITEMS = HALO_P * 8;      // 8-channel chunks
      constexpr int XITER = (ITEMS + 255) / 256;
      // Blocks of 4 load-then-write iterations: full two-phase staging
      // held 48+ loads in registers concurrently and spilled at the
      // 3-waves/SIMD VGPR budget; 4-deep blocks keep the overlapped-
      // latency benefit at 16 VGPRs of staging registers.
      constexpr int XBLK = XITER > 4 ? 4 : XITER;
#pragma unroll
      for (int t0 = 0; t0 < XITER; t0 += XBLK) {
        w4bf16x8 vx[XBLK];
#pragma unroll
        for (int u = 0; u < XBLK; ++u) {
          const int i = tid + (t0 + u) * 256;
          const int rest = i >> 3;
          const int chunk = ((rest & 3) << 1) | (i & 1);
          const int ph = ((rest >> 2) << 2) | ((i >> 1) & 3);
          vx[u] = w4bf16x8{};
          if (i < ITEMS) {
            const int iy = oh0 - pad + ph / HALO_W;
            const int ix = ow0 - pad + ph % HALO_W;
            if (iy >= 0 && iy < H && ix >= 0 && ix < W) {
              vx[u] = *reinterpret_cast<const w4bf16x8*>(
                  x + (((long)img * H + iy) * W + ix) * C + chunk * 8);
            }
          }
        }
#pragma unroll
        for (int u = 0; u < XBLK; ++u) {
          const int i = tid + (t0 + u) * 256;
          if (i < ITEMS) {
            const int rest = i >> 3;
            const int chunk = ((rest & 3) << 1) | (i & 1);
            const int ph = ((rest >> 2) << 2) | ((i >> 1) & 3);
            const int el = ((ph >> 2) * 4 + (chunk >> 1)) * 64 +
                           (ph & 3) * 16 + (chunk & 1) * 8;
            *reinterpret_cast<w4bf16x8*>(&lds_x[el]) = vx[u];
          }
        }
      }
    }
    // ---- stage dy window transposed: lds_dyt[k][win_p] (v3 code) ----
    {
      constexpr int kchunks = K >> 3;
      constexpr int YITER = (WIN_P * kchunks + 255) / 256;
      constexpr int YBLK = YITER > 2 ? 2 : YITER;
#pragma unroll
      for (int t0 = 0; t0 < YITER; t0 += YBLK) {
        w4bf16x8 vy[YBLK];
#pragma unroll
        for (int u = 0; u < YBLK; ++u) {
          const int i = tid + (t0 + u) * 256;
          const int chunk = i % kchunks;
          const int p = i / kchunks;
          const int orow = oh0 + p / W4_TILE_W;
          const int ocol = ow0 + p % W4_TILE_W;
          vy[u] = w4bf16x8{};
          if (i < WIN_P * kchunks && orow < OH && ocol < OW) {
            vy[u] = *reinterpret_cast<const w4bf16x8*>(
                dy + (((long)img * OH + orow) * OW + ocol) * K +
                chunk * 8);
          }
        }
#pragma unroll
        for (int u = 0; u < YBLK; ++u) {
          const int i = tid + (t0 + u) * 256;
          if (i < WIN_P * kchunks) {
            const int chunk = i % kchunks;
            const int p = i / kchunks;
#pragma unroll
            for (int j = 0; j < 8; ++j)
              lds_dyt[(chunk * 8 + j) * DYT_P + p] = vy[u][j];
          }
        }
      }
    }
    __syncthreads();

    {
      const int k = ntile * 32 + mrow;
#pragma unroll 2
      for (int kstep = 0; kstep < WIN_P / 16; ++kstep) {
        w4bf16x8 b_frag = *reinterpret_cast<const w4bf16x8*>(
            &lds_dyt[k * DYT_P + kstep * 16 + kgrp * 8]);
#pragma unroll
        for (int g = 0; g < W4_GROUP; ++g) {
          const int rs = rs_base + g;
          if (rs < RS) {
            const int r = rs / S, s = rs % S;
            // Contraction pixels for (kstep, r, s) are the 16
            // CONSECUTIVE packed halo pixels starting at
            // (kstep+r)*HALO_W + s; this lane reads its 8 via two
            // tr_b16 (pixels +0..3 and +4..7 = +512 B).
            const int pl = (kstep + r) * HALO_W + s + pix_lane;
            const int el = ((pl >> 2) * 4 + (cpiece >> 4)) * 64 +
                           (pl & 3) * 16 + (cpiece & 15);
            unsigned addr = (unsigned)(unsigned long long)
                (__attribute__((address_space(3))) short*)&lds_x[el];
            w4short4 lo, hi;
            asm volatile(
                "ds_read_b64_tr_b16 %0, %2\n\t"
                "ds_read_b64_tr_b16 %1, %2 offset:512\n\t"
                "s_waitcnt lgkmcnt(0)"
                : "=&v"(lo), "=&v"(hi) : "v"(addr));
            const w4bf16x8 a_frag = __builtin_shufflevector(
                lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
            acc[g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a_frag, b_frag, acc[g], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  float* part = dw_part
      + ((long)blockIdx.x * gridDim.y + blockIdx.y) * W4_GROUP * C * K;
#pragma unroll
  for (int g = 0; g < W4_GROUP; ++g) {
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

__global__ void __launch_bounds__(256)
wrw4_reduce_kernel(const float* __restrict__ part,
                   float* __restrict__ dw, int RS, long ck,
                   int ngx, int ngy) {
  const long cells = (long)RS * ck;
  for (long i = blockIdx.x * 256L + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int rs = (int)(i / ck);
    const long rest = i % ck;
    const int gy = rs / W4_GROUP, g = rs % W4_GROUP;
    float s = 0.0f;
    for (int gx = 0; gx < ngx; ++gx) {
      s += part[(((long)gx * ngy + gy) * W4_GROUP + g) * ck + rest];
    }
    dw[i] = s;
  }
}

at::Tensor conv_s1_wrw4(at::Tensor x, at::Tensor dy, int64_t R,
                        int64_t S, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_s1_wrw4: bf16 x required");
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "conv_s1_wrw4: bf16 dy required");
  x = x.contiguous(at::MemoryFormat::ChannelsLast);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(C == 64 && K == 64, "conv_s1_wrw4: C == K == 64 only");
  TORCH_CHECK(OH == H + 2 * pad - R + 1 && OW == W + 2 * pad - S + 1,
              "conv_s1_wrw4: dy shape mismatch");
  const int tiles_h = (OH + 7) / 8;
  const int tiles_w = (OW + W4_TILE_W - 1) / W4_TILE_W;
  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int ngy = (int)((R * S + W4_GROUP - 1) / W4_GROUP);
  const int window_groups = std::min(total_windows, 768 / ngy);
  auto part = at::empty({(long)window_groups * ngy, W4_GROUP,
                         (long)C, K}, x.options().dtype(at::kFloat));
  auto dw = at::empty({(long)R * S, C, K},
                      x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(window_groups, ngy), dim3(256),
                       0, stream.stream(),
                       (const w4bf16_t*)x.data_ptr(),
                       (const w4bf16_t*)dy.data_ptr(),
                       (float*)part.data_ptr(),
                       N, H, W, (int)pad, OH, OW, tiles_h, tiles_w,
                       window_groups);
  };
  if (R == 3 && S == 3) launch(conv_s1_wrw4_kernel<3, 3>);
  else if (R == 5 && S == 5) launch(conv_s1_wrw4_kernel<5, 5>);
  else TORCH_CHECK(false, "conv_s1_wrw4: R/S must be 3x3 or 5x5");
  const long cells = (long)R * S * C * K;
  const int rblocks = (int)std::min((cells + 255) / 256, 1024L);
  hipLaunchKernelGGL(wrw4_reduce_kernel, dim3(rblocks), dim3(256), 0,
                     stream.stream(), (const float*)part.data_ptr(),
                     (float*)dw.data_ptr(), (int)(R * S), (long)C * K,
                     window_groups, ngy);
  return dw;
}
