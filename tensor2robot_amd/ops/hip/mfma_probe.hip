// MFMA fragment-layout probe for v_mfma_f32_32x32x16_bf16 (gfx950).
//
// Computes one 32x32 tile D = A(32x16) @ B(16x32) with the assumed
// per-lane fragment layouts; the GPU test compares against torch.matmul
// with ASYMMETRIC inputs (guide §3: symmetric B hides row/col swaps).
//
// Assumed layouts (verified by tests/test_mfma_gpu.py):
//   A: lane l holds 8 bf16, row i = l % 32, k = (l>>5)*8 + j
//   B: lane l holds 8 bf16, col n = l % 32, k = (l>>5)*8 + j
//   C/D: 16 f32/lane, col = l & 31, row = (reg&3) + 8*(reg>>2) + 4*(l>>5)
//        (cdna_hip_programming.md §3 fragment layout)

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef __hip_bfloat16 pbf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

extern "C" __global__ void __launch_bounds__(64)
mfma_probe_kernel(const pbf16_t* __restrict__ A,  // [32][16] row-major
                  const pbf16_t* __restrict__ B,  // [16][32] row-major
                  float* __restrict__ D) {        // [32][32] row-major
  const int l = threadIdx.x;
  bf16x8 a_frag, b_frag;
  const int row = l % 32;
  const int kbase = (l >> 5) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a_frag[j] = *reinterpret_cast<const short*>(&A[row * 16 + kbase + j]);
    b_frag[j] = *reinterpret_cast<const short*>(&B[(kbase + j) * 32 + row]);
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag, acc, 0, 0,
                                                0);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int out_row = (reg & 3) + 8 * (reg >> 2) + 4 * (l >> 5);
    const int out_col = l & 31;
    D[out_row * 32 + out_col] = acc[reg];
  }
}

at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}));
  TORCH_CHECK(B.sizes() == at::IntArrayRef({16, 32}));
  auto D = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     stream.stream(),
                     (const pbf16_t*)A.contiguous().data_ptr(),
                     (const pbf16_t*)B.contiguous().data_ptr(),
                     (float*)D.data_ptr());
  return D;
}
