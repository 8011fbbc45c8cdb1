// MFMA layout probe: one v_mfma_f32_16x16x32_bf16 tile, used by the GPU test
// suite to pin the exact lane->element mapping the conv/GEMM kernels assume
// (guide §3: always verify with ASYMMETRIC operands).
//
// Assumed mapping (CDNA4 16x16x32 bf16):
//   A (16M x 32K): lane l holds A[l & 15][(l >> 4)*8 + j], j = 0..7
//   B (32K x 16N): lane l holds B[(l >> 4)*8 + j][l & 15]
//   C/D (16x16 f32): lane l holds C[(l >> 4)*4 + v][l & 15], v = 0..3

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,  // (16,32) row-major
                                  const __bf16* __restrict__ B,  // (32,16) row-major
                                  float* __restrict__ C) {       // (16,16) row-major
  const int l = threadIdx.x & 63;
  bf16x8 a, b;
  const int ar = l & 15;
  const int ak = (l >> 4) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[ar * 32 + ak + j];
    b[j] = B[(ak + j) * 16 + (l & 15)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int v = 0; v < 4; ++v) {
    C[((l >> 4) * 4 + v) * 16 + (l & 15)] = c[v];
  }
}

}  // namespace

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  CHECK_INPUT(A);
  CHECK_INPUT(B);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 && A.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(B.scalar_type() == torch::kBFloat16 && B.sizes() == torch::IntArrayRef({32, 16}));
  auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const __bf16*>(A.data_ptr()),
                     reinterpret_cast<const __bf16*>(B.data_ptr()),
                     C.data_ptr<float>());
  return C;
}
