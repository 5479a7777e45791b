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
//  - The rs grid dimension is aligned to WEIGHT ROWS: blockIdx.y = r,
//    the in-wave accumulator group walks s = 0..S-1.  With r fixed,
//    output-tile row kstep only ever contracts against x row
//    (kstep + r), so the WG stages just the 8 x rows it needs
//    (8 x 20 pixels = 20.5 KiB) instead of the full halo.
//  - x is staged PIXEL-major in packed [ph/4][c/16][4][16] subtiles
//    (one 16-B vector write per 8-channel chunk, no transpose
//    scatter).  The row pitch is padded to 20 pixels for both kernel
//    sizes, which makes the packed pixel index of a contraction slice
//    pl = kstep*20 + s + lane_pixel: 20 = 0 (mod 4), so the LDS byte
//    address decomposes as  addr_s[lane]  +  kstep*2560  — five
//    precomputed per-lane address registers and a compile-time
//    offset immediate; the MFMA inner loop has ZERO address VALU.
//  - Each A fragment is TWO ds_read_b64_tr_b16: per 16-lane group the
//    instruction reads a [4 pixel][16 channel] row-major block and
//    delivers lane (l&15) its channel-column of 4 pixels; pixels
//    +4..7 live exactly +512 B away (next subtile block), so the
//    second read is the same address register with offset +512.
//    Every lane address is 8-B aligned by construction (the G17
//    tr_b16 trap: a base = 2/4/6 mod 8 shorts silently returns the
//    8-aligned address's data).  The 16 lane addresses of a group
//    cover 32 distinct banks for EVERY s shift (pixel row = 32 B,
//    block = 128 B), i.e. a conflict-free tr subtiling.
//
// dy stays in the v3 transposed image read with aligned b128 (its
// fragments are always 16-B aligned).  Flush follows v3 (non-atomic
// per-WG partials + reduce kernel).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 w4bf16_t;
typedef __attribute__((ext_vector_type(8))) short w4bf16x8;
typedef __attribute__((ext_vector_type(4))) short w4short4;
typedef __attribute__((ext_vector_type(16))) float w4f32x16;

#define W4_TILE_W 16
#define W4_ROW_W 20            // padded x row pitch (pixels), % 4 == 0

// 3x3 (48 acc VGPRs) holds 4 waves/SIMD spill-free; 5x5 (80 acc) needs
// the 3-wave VGPR budget to stay spill-free — 10 inner-loop spills at
// the 4-wave cap cost more than the lost wave.
template <int R, int S>
__global__ void __launch_bounds__(256, (R * S > 9) ? 3 : 4)
conv_s1_wrw4_kernel(const w4bf16_t* __restrict__ x,
                    const w4bf16_t* __restrict__ dy,
                    float* __restrict__ dw_part,  // [gx][r][S][C][K]
                    int N, int H, int W, int pad,
                    int OH, int OW, int tiles_h, int tiles_w,
                    int window_groups) {
  constexpr int C = 64, K = 64;
  constexpr int TH = 8;
  constexpr int WIN_P = TH * W4_TILE_W;          // 128
  constexpr int HALO_W = W4_TILE_W + S - 1;      // cols actually used
  constexpr int XROWS = TH;                      // one x row per kstep
  constexpr int X_P = XROWS * W4_ROW_W;          // 160 pixels
  constexpr int DYT_P = WIN_P + 8;
  // x image: element (ph, c) lives at
  //   ((ph>>2)*4 + (c>>4))*64 + (ph&3)*16 + (c&15).
  // 16-B aligned base: tr_b16 lane addresses derive from it and a base
  // = 2/4/6 mod 8 shorts silently reads wrong data (guide G17).
  __shared__ __attribute__((aligned(16))) short lds_x[X_P * C];
  __shared__ __attribute__((aligned(16))) short lds_dyt[K * DYT_P];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int mtile = wave & 1;              // c half
  const int ntile = wave >> 1;             // k half
  const int r_fixed = blockIdx.y;          // weight row this WG owns

  w4f32x16 acc[S];
#pragma unroll
  for (int g = 0; g < S; ++g) acc[g] = w4f32x16{};

  const int total_windows = (int)((long)N * tiles_h * tiles_w);
  const int mrow = lane & 31;
  const int kgrp = lane >> 5;

  // tr_b16 per-lane addresses, one per s shift — constant across the
  // window loop (the LDS image location is fixed).  Lane l of 16-lane
  // group grp covers pixel pl = kstep*20 + s + (grp>>1)*8 + ((l&15)>>2)
  // at channel piece mtile*32 + (grp&1)*16 + ((l&15)&3)*4; the kstep
  // part is the offset immediate.
  const int grp = lane >> 4;
  const int m16 = lane & 15;
  const int pix_lane = (grp >> 1) * 8 + (m16 >> 2);
  const int cpiece = mtile * 32 + (grp & 1) * 16 + (m16 & 3) * 4;
  unsigned addr_s[S];
#pragma unroll
  for (int s = 0; s < S; ++s) {
    const int pl = s + pix_lane;
    const int el = ((pl >> 2) * 4 + (cpiece >> 4)) * 64 +
                   (pl & 3) * 16 + (cpiece & 15);
    addr_s[s] = (unsigned)(unsigned long long)
        (__attribute__((address_space(3))) short*)&lds_x[el];
  }

  for (int win = blockIdx.x; win < total_windows; win += window_groups) {
    const int img = win / (tiles_h * tiles_w);
    const int trest = win % (tiles_h * tiles_w);
    const int oh0 = (trest / tiles_w) * TH;
    const int ow0 = (trest % tiles_w) * W4_TILE_W;

    // ---- stage the 8 x rows this r needs, pixel-major.  Lane->(ph,
    // chunk) mapping staggers (ph&3, chunk&1) across each 8-lane
    // write-service group so the b128 writes are bank-conflict-free.
    {
      constexpr int ITEMS = X_P * 8;         // 8-channel chunks
      constexpr int XITER = (ITEMS + 255) / 256;   // 5
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
            const int hcol = ph % W4_ROW_W;
            const int iy = oh0 - pad + r_fixed + ph / W4_ROW_W;
            const int ix = ow0 - pad + hcol;
            if (hcol < HALO_W && iy >= 0 && iy < H &&
                ix >= 0 && ix < W) {
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
#pragma unroll
      for (int kstep = 0; kstep < WIN_P / 16; ++kstep) {
        w4bf16x8 b_frag = *reinterpret_cast<const w4bf16x8*>(
            &lds_dyt[k * DYT_P + kstep * 16 + kgrp * 8]);
        // All 2S tr reads of this kstep issue back-to-back in ONE asm
        // block with a single waitcnt: LDS latency is paid once per
        // kstep (reads pipeline through the LDS) instead of once per
        // s.  kstep lives in the compile-time offset immediate:
        // +kstep*20 pixels = +kstep*5 blocks = +kstep*2560 bytes.
        w4short4 fr[2 * S];
#define W4_CASE3(KS) \
          case KS: \
            asm volatile( \
                "ds_read_b64_tr_b16 %0, %6 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %1, %6 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %2, %7 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %3, %7 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %4, %8 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %5, %8 offset:" #KS "*2560+512\n\t" \
                "s_waitcnt lgkmcnt(0)" \
                : "=&v"(fr[0]), "=&v"(fr[1]), "=&v"(fr[2]), \
                  "=&v"(fr[3]), "=&v"(fr[4]), "=&v"(fr[5]) \
                : "v"(addr_s[0]), "v"(addr_s[1]), "v"(addr_s[2])); \
            break;
#define W4_CASE5(KS) \
          case KS: \
            asm volatile( \
                "ds_read_b64_tr_b16 %0, %10 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %1, %10 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %2, %11 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %3, %11 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %4, %12 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %5, %12 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %6, %13 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %7, %13 offset:" #KS "*2560+512\n\t" \
                "ds_read_b64_tr_b16 %8, %14 offset:" #KS "*2560\n\t" \
                "ds_read_b64_tr_b16 %9, %14 offset:" #KS "*2560+512\n\t" \
                "s_waitcnt lgkmcnt(0)" \
                : "=&v"(fr[0]), "=&v"(fr[1]), "=&v"(fr[2]), \
                  "=&v"(fr[3]), "=&v"(fr[4]), "=&v"(fr[5]), \
                  "=&v"(fr[6]), "=&v"(fr[7]), "=&v"(fr[8]), \
                  "=&v"(fr[9]) \
                : "v"(addr_s[0]), "v"(addr_s[1]), "v"(addr_s[2]), \
                  "v"(addr_s[3]), "v"(addr_s[4])); \
            break;
#define W4_CASES(M) M(0) M(1) M(2) M(3) M(4) M(5) M(6) M(7)
        if constexpr (S == 3) {
          switch (kstep) { W4_CASES(W4_CASE3) }
        } else {
          switch (kstep) { W4_CASES(W4_CASE5) }
        }
#undef W4_CASES
#undef W4_CASE3
#undef W4_CASE5
#pragma unroll
        for (int g = 0; g < S; ++g) {
          const w4bf16x8 a_frag = __builtin_shufflevector(
              fr[2 * g], fr[2 * g + 1], 0, 1, 2, 3, 4, 5, 6, 7);
          acc[g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_frag, b_frag, acc[g], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  float* part = dw_part
      + ((long)blockIdx.x * gridDim.y + blockIdx.y) * S * C * K;
#pragma unroll
  for (int g = 0; g < S; ++g) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int c = mtile * 32
          + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int k = ntile * 32 + (lane & 31);
      part[((long)g * C + c) * K + k] = acc[g][reg];
    }
  }
}

// Reduce: dw[k][c][r][s] (bf16, the autograd-facing [K,C,R,S] weight
// gradient — layout permute and bf16 cast fused here instead of two
// extra torch dispatches per conv) = sum over gx of
// part[gx][rs/S][rs%S][c][k].  Loop order follows the partials (rs,
// c, k fastest) so the ngx reads per cell stay coalesced; the single
// output write scatters.
__global__ void __launch_bounds__(256)
wrw4_reduce_kernel(const float* __restrict__ part,
                   w4bf16_t* __restrict__ dw, int RS, long ck,
                   int ngx, int ngy, int group, int C) {
  const long cells = (long)RS * ck;
  const long gx_stride = (long)ngy * group * ck;
  for (long i = blockIdx.x * 256L + threadIdx.x; i < cells;
       i += (long)gridDim.x * 256) {
    const int rs = (int)(i / ck);
    const long rest = i % ck;
    const int gy = rs / group, g = rs % group;
    const float* base = part + ((long)gy * group + g) * ck + rest;
    // 4 accumulator chains: the single-chain version was latency-bound
    // (400 WGs x ~150 dependent adds measured 38 us for a 63 MB read).
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    int gx = 0;
    for (; gx + 4 <= ngx; gx += 4) {
      s0 += base[(gx + 0) * gx_stride];
      s1 += base[(gx + 1) * gx_stride];
      s2 += base[(gx + 2) * gx_stride];
      s3 += base[(gx + 3) * gx_stride];
    }
    for (; gx < ngx; ++gx) s0 += base[gx * gx_stride];
    const int c = (int)(rest / (ck / C));
    const int k = (int)(rest % (ck / C));
    dw[((long)k * C + c) * RS + rs] = (w4bf16_t)(s0 + s1 + s2 + s3);
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
  const int ngy = (int)R;
  // Fill EXACTLY one residency wave: 3 WGs/CU for the 5x5 (VGPR
  // budget), 4 for the 3x3 -> 768/1024 concurrent WGs.  A partial
  // second wave of straggler WGs costs ~18% (measured 0.137 -> 0.162
  // ms when the grid was 1020 WGs at 768 slots).
  const int resident = (R * S > 9) ? 768 : 1024;
  const int window_groups = std::min(total_windows,
                                     std::max(1, resident / ngy));
  auto part = at::empty({(long)window_groups * ngy, S,
                         (long)C, K}, x.options().dtype(at::kFloat));
  auto dw = at::empty({(long)K, C, R, S},
                      x.options().dtype(at::kBFloat16));
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
                     (w4bf16_t*)dw.data_ptr(), (int)(R * S),
                     (long)C * K, window_groups, ngy, (int)S, (int)C);
  return dw;
}
