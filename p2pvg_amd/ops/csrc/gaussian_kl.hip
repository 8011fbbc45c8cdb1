// Fused closed-form gaussian KL.
//
// KL(N(mu1, e^lv1) || N(mu2, e^lv2)) summed over all elements / denom, with
// analytic backward. The reference computes this as ~8 ATen elementwise
// kernels + a reduction per timestep (reference misc/criterion.py:10-15);
// here it is one kernel each way on (B, z_dim) tensors (~1K elements:
// pure launch-bound, the fusion IS the win).

#include "common.h"

namespace {

__global__ void gaussian_kl_fwd_kernel(const float* __restrict__ mu1,
                                       const float* __restrict__ lv1,
                                       const float* __restrict__ mu2,
                                       const float* __restrict__ lv2,
                                       float* __restrict__ rows,  // per-block
                                       float inv_denom, long n) {
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float dmu = mu1[i] - mu2[i];
    acc += 0.5f * (lv2[i] - lv1[i]) +
           (__expf(lv1[i]) + dmu * dmu) / (2.f * __expf(lv2[i])) - 0.5f;
  }
  // deterministic: wave shfl reduce -> per-wave LDS slot -> serial combine
  // by thread 0 -> per-block row store (scalar_rows_sum finishes serially)
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  __shared__ float sw[256 / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) sw[threadIdx.x / WAVE] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < 256 / WAVE; ++w) t += sw[w];
    rows[blockIdx.x] = t * inv_denom;
  }
}

// fixed-structure (deterministic) 256-thread strided sum + LDS tree; the
// single-thread serial walk cost ~59 us at n=1280 from the dependent-load
// chain — this is ~3 us.
__global__ void scalar_rows_sum_kernel(const float* __restrict__ rows, int n,
                                       float* __restrict__ out) {
  float t = 0.f;
  for (int i = threadIdx.x; i < n; i += 256) t += rows[i];
  __shared__ float sw[256];
  sw[threadIdx.x] = t;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if ((int)threadIdx.x < off) sw[threadIdx.x] += sw[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[0] = sw[0];
}

__global__ void gaussian_kl_bwd_kernel(
    const float* __restrict__ mu1, const float* __restrict__ lv1,
    const float* __restrict__ mu2, const float* __restrict__ lv2,
    const float* __restrict__ dout,  // scalar
    float* __restrict__ dmu1, float* __restrict__ dlv1,
    float* __restrict__ dmu2, float* __restrict__ dlv2, float inv_denom,
    long n) {
  const float s = dout[0] * inv_denom;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float e1 = __expf(lv1[i]);
    const float inv_2e2 = 0.5f * __expf(-lv2[i]);
    const float dmu = mu1[i] - mu2[i];
    // d/dmu1 = (mu1-mu2)/e^lv2 ; d/dmu2 = -that
    const float g_mu = 2.f * dmu * inv_2e2;
    dmu1[i] = s * g_mu;
    dmu2[i] = -s * g_mu;
    // d/dlv1 = -1/2 + e^lv1/(2 e^lv2)
    dlv1[i] = s * (-0.5f + e1 * inv_2e2);
    // d/dlv2 = 1/2 - (e^lv1 + dmu^2)/(2 e^lv2)
    dlv2[i] = s * (0.5f - (e1 + dmu * dmu) * inv_2e2);
  }
}

}  // namespace

torch::Tensor gaussian_kl_fwd(torch::Tensor mu1, torch::Tensor lv1,
                              torch::Tensor mu2, torch::Tensor lv2,
                              double denom) {
  CHECK_INPUT(mu1);
  CHECK_INPUT(lv1);
  CHECK_INPUT(mu2);
  CHECK_INPUT(lv2);
  auto out = torch::empty({}, mu1.options());
  const long n = mu1.numel();
  const int threads = 256;
  const int blocks =
      (int)std::max<long>(1, std::min<long>(64, (n + threads - 1) / threads));
  auto rows = torch::empty({blocks}, mu1.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gaussian_kl_fwd_kernel, dim3(blocks),
                     dim3(threads), 0, stream, mu1.data_ptr<float>(),
                     lv1.data_ptr<float>(), mu2.data_ptr<float>(),
                     lv2.data_ptr<float>(), rows.data_ptr<float>(),
                     (float)(1.0 / denom), n);
  hipLaunchKernelGGL(scalar_rows_sum_kernel, dim3(1), dim3(256), 0, stream,
                     rows.data_ptr<float>(), blocks, out.data_ptr<float>());
  return out;
}

std::vector<torch::Tensor> gaussian_kl_bwd(torch::Tensor mu1, torch::Tensor lv1,
                                           torch::Tensor mu2, torch::Tensor lv2,
                                           torch::Tensor dout, double denom) {
  auto dmu1 = torch::empty_like(mu1);
  auto dlv1 = torch::empty_like(lv1);
  auto dmu2 = torch::empty_like(mu2);
  auto dlv2 = torch::empty_like(lv2);
  const long n = mu1.numel();
  const int threads = 256;
  const int blocks = std::min<long>(64, (n + threads - 1) / threads);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gaussian_kl_bwd_kernel, dim3(std::max(blocks, 1)),
                     dim3(threads), 0, stream, mu1.data_ptr<float>(),
                     lv1.data_ptr<float>(), mu2.data_ptr<float>(),
                     lv2.data_ptr<float>(), dout.data_ptr<float>(),
                     dmu1.data_ptr<float>(), dlv1.data_ptr<float>(),
                     dmu2.data_ptr<float>(), dlv2.data_ptr<float>(),
                     (float)(1.0 / denom), n);
  return {dmu1, dlv1, dmu2, dlv2};
}

// ---------------------------------------------------------------------------
// Fused squared-difference sum (the MSE reductions).
// ATen's mse_loss on the (B,C,64,64) bf16 frame tensors dispatches a
// ReduceOp<BFloat16> measured at ~320us per call (~25x off bandwidth);
// this kernel reads both tensors once, accumulates fp32 in registers,
// wave-shfl reduces and lands ONE atomic per block.

namespace {

typedef __bf16 bf16v8 __attribute__((ext_vector_type(8)));
typedef float f32v8 __attribute__((ext_vector_type(8)));

template <typename VB, typename TB>
__global__ __launch_bounds__(256) void sqdiff_sum_kernel(
    const __bf16* __restrict__ a, const TB* __restrict__ b,
    float* __restrict__ out, long nvec) {
  float acc = 0.f;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < nvec;
       i += (long)gridDim.x * 256) {
    bf16v8 av = *reinterpret_cast<const bf16v8*>(a + i * 8);
    VB bv = *reinterpret_cast<const VB*>(b + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float d = (float)av[j] - (float)bv[j];
      acc += d * d;
    }
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  __shared__ float sw[256 / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) sw[threadIdx.x / WAVE] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < 256 / WAVE; ++w) t += sw[w];
    out[blockIdx.x] = t;   // per-block row; serial combine follows
  }
}

__global__ void sq_rows_sum_kernel(const float* __restrict__ rows, int n,
                                   float* __restrict__ out) {
  float t = 0.f;
  for (int i = threadIdx.x; i < n; i += 256) t += rows[i];
  __shared__ float sw[256];
  sw[threadIdx.x] = t;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if ((int)threadIdx.x < off) sw[threadIdx.x] += sw[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[0] = sw[0];
}

}  // namespace

// sum((a-b)^2) over all elements; a bf16, b bf16 or fp32 (no cast pass),
// identical sizes/strides (dense), numel % 8 == 0. fp32 scalar out.
torch::Tensor sqdiff_sum(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16,
              "sqdiff_sum: a must be bf16 CUDA");
  TORCH_CHECK(a.sizes() == b.sizes() && a.strides() == b.strides() &&
              a.is_non_overlapping_and_dense(),
              "sqdiff_sum: layouts must match and be dense");
  TORCH_CHECK(a.numel() % 8 == 0, "sqdiff_sum: numel % 8 != 0");
  auto out = torch::empty({}, a.options().dtype(torch::kFloat32));
  const long nvec = a.numel() / 8;
  const int grid = (int)std::max<long>(1, std::min<long>(1280, (nvec + 255) / 256));
  auto rows = torch::empty({grid}, a.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (b.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((sqdiff_sum_kernel<bf16v8, __bf16>), dim3(grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const __bf16*>(a.data_ptr()),
                       reinterpret_cast<const __bf16*>(b.data_ptr()),
                       rows.data_ptr<float>(), nvec);
  } else {
    TORCH_CHECK(b.scalar_type() == torch::kFloat32,
                "sqdiff_sum: b must be bf16 or fp32");
    hipLaunchKernelGGL((sqdiff_sum_kernel<f32v8, float>), dim3(grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const __bf16*>(a.data_ptr()),
                       b.data_ptr<float>(), rows.data_ptr<float>(), nvec);
  }
  hipLaunchKernelGGL(sq_rows_sum_kernel, dim3(1), dim3(256), 0, stream,
                     rows.data_ptr<float>(), grid, out.data_ptr<float>());
  return out;
}
