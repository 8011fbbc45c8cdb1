// Multi-tensor fused Adam for gfx950.
//
// The reference steps five independent torch.optim.Adam instances
// (reference models/p2p_model.py:51-57,273-287), each a pile of per-tensor
// kernels. Here one launch updates up to CHUNK_TENSORS tensors: param, grad,
// exp_avg, exp_avg_sq pointers ride in the kernarg block; blocks grid-stride
// within (tensor, chunk) space. fp32, bias-corrected, matching
// torch.optim.Adam exactly.
//
// The step count is read from DEVICE memory (one fp32 scalar, incremented by
// the caller with a tensor op) so the whole optimizer step is
// hipGraph-capturable: no host-side state is baked into the launch.

#include "common.h"

namespace {

constexpr int CHUNK_TENSORS = 48;
constexpr int BLOCK = 256;
constexpr int ILP = 4;

struct AdamArgs {
  float* p[CHUNK_TENSORS];
  float* g[CHUNK_TENSORS];
  float* m[CHUNK_TENSORS];
  float* v[CHUNK_TENSORS];
  long n[CHUNK_TENSORS];
};

__global__ __launch_bounds__(BLOCK) void adam_kernel(
    AdamArgs args, int ntensors, float lr, float beta1, float beta2, float eps,
    float weight_decay, const float* __restrict__ step_ptr) {
  const int t = blockIdx.y;
  if (t >= ntensors) return;
  float* __restrict__ p = args.p[t];
  float* __restrict__ g = args.g[t];
  float* __restrict__ m = args.m[t];
  float* __restrict__ v = args.v[t];
  const long n = args.n[t];

  const float step = step_ptr[0];
  const float bc1 = 1.f - __powf(beta1, step);
  const float bc2 = 1.f - __powf(beta2, step);
  const float step_size = lr / bc1;
  const float inv_sqrt_bc2 = rsqrtf(bc2);

  for (long i = (long)blockIdx.x * BLOCK * ILP + threadIdx.x * ILP; i < n;
       i += (long)gridDim.x * BLOCK * ILP) {
#pragma unroll
    for (int k = 0; k < ILP; ++k) {
      const long idx = i + k;
      if (idx < n) {
        float gv = g[idx];
        if (weight_decay != 0.f) gv += weight_decay * p[idx];
        const float mv = beta1 * m[idx] + (1.f - beta1) * gv;
        const float vv = beta2 * v[idx] + (1.f - beta2) * gv * gv;
        m[idx] = mv;
        v[idx] = vv;
        // denom = sqrt(v)/sqrt(bc2) + eps; p -= lr/bc1 * m / denom
        p[idx] -= step_size * mv / (sqrtf(vv) * inv_sqrt_bc2 + eps);
      }
    }
  }
}

}  // namespace

void multi_tensor_adam(std::vector<torch::Tensor> params,
                       std::vector<torch::Tensor> grads,
                       std::vector<torch::Tensor> exp_avgs,
                       std::vector<torch::Tensor> exp_avg_sqs, double lr,
                       double beta1, double beta2, double eps,
                       double weight_decay, torch::Tensor step) {
  TORCH_CHECK(params.size() == grads.size() && params.size() == exp_avgs.size() &&
                  params.size() == exp_avg_sqs.size(),
              "multi_tensor_adam: list length mismatch");
  CHECK_CUDA(step);
  TORCH_CHECK(step.scalar_type() == torch::kFloat32,
              "multi_tensor_adam: step must be a device fp32 scalar");
  auto stream = at::cuda::getCurrentCUDAStream();

  size_t i = 0;
  while (i < params.size()) {
    AdamArgs args;
    int nt = 0;
    long max_n = 0;
    for (; nt < CHUNK_TENSORS && i < params.size(); ++nt, ++i) {
      // elementwise update: any dense layout is fine (channels_last conv
      // weights included) as long as param/grad/state share it
      CHECK_CUDA(params[i]);
      TORCH_CHECK(params[i].is_non_overlapping_and_dense(),
                  "multi_tensor_adam: param must be dense");
      TORCH_CHECK(grads[i].strides() == params[i].strides(),
                  "multi_tensor_adam: grad layout must match param");
      TORCH_CHECK(exp_avgs[i].strides() == params[i].strides() &&
                      exp_avg_sqs[i].strides() == params[i].strides(),
                  "multi_tensor_adam: state layout must match param");
      TORCH_CHECK(params[i].scalar_type() == torch::kFloat32,
                  "multi_tensor_adam: fp32 only");
      args.p[nt] = params[i].data_ptr<float>();
      args.g[nt] = grads[i].data_ptr<float>();
      args.m[nt] = exp_avgs[i].data_ptr<float>();
      args.v[nt] = exp_avg_sqs[i].data_ptr<float>();
      args.n[nt] = params[i].numel();
      max_n = std::max(max_n, args.n[nt]);
    }
    const int bx = std::min<long>(256, (max_n + BLOCK * ILP - 1) / (BLOCK * ILP));
    dim3 grid(std::max(bx, 1), nt);
    hipLaunchKernelGGL(adam_kernel, grid, dim3(BLOCK), 0, stream, args, nt,
                       (float)lr, (float)beta1, (float)beta2, (float)eps,
                       (float)weight_decay, step.data_ptr<float>());
  }
}
