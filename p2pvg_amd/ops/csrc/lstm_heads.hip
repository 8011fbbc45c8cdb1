// Fused LSTM stack heads for gfx950.
//
// The reference assembles every LSTM input with torch.cat and runs the
// embed / mu / logvar / output Linears as separate GEMMs
// (reference models/lstm.py:13,16,56-57, models/p2p_model.py:241-247) —
// ~20 small kernels per timestep on (B,258)x(258,256)-class shapes that are
// LATENCY-bound (weights are L2-resident, the graph replay gap dominates).
// Here each head is ONE kernel each way:
//
// - affine4_fwd: out = [h | g | s1 | s2] @ W^T + b WITHOUT materializing the
//   concat (the virtual K columns gather from the four sources directly).
// - gauss_head_fwd: mu = hW_m^T + b_m, lv = hW_l^T + b_l,
//   z = eps * exp(lv/2) + mu in one pass (eps supplied by the caller's
//   philox draw so hipGraph RNG semantics are untouched).
// - tanh_head_fwd: out = tanh(hW^T + b).
//
// Backwards accumulate dW/db STRAIGHT into the managed fp32 .grad buffers
// (same no-AccumulateGrad flow as the conv kernels); each (n,k) cell is
// owned by exactly one thread so the accumulation is deterministic.
// All fp32 (the recurrent state path stays fp32 under autocast).

#include "common.h"

namespace {

constexpr int BLK = 256;

// out[r][n] = b[n] + sum_k W[n][k] * src(r,k) where the virtual source is
// [h (K1) | g (K2) | s1 | s2]; thread per output element, K-serial (K<=258:
// latency-class kernel, weights L2-resident).
__global__ __launch_bounds__(BLK) void affine4_fwd_kernel(
    const float* __restrict__ h, const float* __restrict__ g,
    const float* __restrict__ s1, const float* __restrict__ s2,
    const float* __restrict__ W, const float* __restrict__ b,
    float* __restrict__ out, int B, int N, int K1, int K2) {
  const int K = K1 + K2 + 2;
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)B * N;
       i += (long)gridDim.x * BLK) {
    const int r = (int)(i / N);
    const int n = (int)(i - (long)r * N);
    const float* w = W + (long)n * K;
    const float* hr = h + (long)r * K1;
    const float* gr = g + (long)r * K2;
    float acc = b != nullptr ? b[n] : 0.f;
    for (int k = 0; k < K1; ++k) acc += w[k] * hr[k];
    for (int k = 0; k < K2; ++k) acc += w[K1 + k] * gr[k];
    acc += w[K1 + K2] * s1[r] + w[K1 + K2 + 1] * s2[r];
    out[i] = acc;
  }
}

// dsrc[r][k] = sum_n gout[r][n] * W[n][k + off] for one dense source part
__global__ __launch_bounds__(BLK) void affine_dsrc_kernel(
    const float* __restrict__ gout, const float* __restrict__ W,
    float* __restrict__ dsrc, int B, int N, int K, int Kpart, int off) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)B * Kpart;
       i += (long)gridDim.x * BLK) {
    const int r = (int)(i / Kpart);
    const int k = (int)(i - (long)r * Kpart);
    const float* go = gout + (long)r * N;
    const float* w = W + off + k;
    float acc = 0.f;
    for (int n = 0; n < N; ++n) acc += go[n] * w[(long)n * K];
    dsrc[i] = acc;
  }
}

// dW[n][k] (+)= sum_r gout[r][n] * src(r,k) over the whole virtual source
// (including the two scalar columns), db[n] (+)= sum_r gout[r][n].
// One thread owns one (n, k) — deterministic, no atomics.
__global__ __launch_bounds__(BLK) void affine4_dw_kernel(
    const float* __restrict__ gout, const float* __restrict__ h,
    const float* __restrict__ g, const float* __restrict__ s1,
    const float* __restrict__ s2, float* __restrict__ dW,
    float* __restrict__ db, int B, int N, int K1, int K2, int accumulate) {
  const int K = K1 + K2 + 2;
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)N * (K + 1);
       i += (long)gridDim.x * BLK) {
    const int n = (int)(i / (K + 1));
    const int k = (int)(i - (long)n * (K + 1));
    float acc = 0.f;
    if (k == K) {             // bias column
      if (db == nullptr) continue;
      for (int r = 0; r < B; ++r) acc += gout[(long)r * N + n];
      db[n] = accumulate ? db[n] + acc : acc;
      continue;
    }
    const float* src;
    int stride, kk;
    if (k < K1) { src = h; stride = K1; kk = k; }
    else if (k < K1 + K2) { src = g; stride = K2; kk = k - K1; }
    else if (k == K1 + K2) { src = s1; stride = 1; kk = 0; }
    else { src = s2; stride = 1; kk = 0; }
    for (int r = 0; r < B; ++r)
      acc += gout[(long)r * N + n] * src[(long)r * stride + kk];
    dW[(long)n * K + k] = accumulate ? dW[(long)n * K + k] + acc : acc;
  }
}

// mu/lv/z in one pass; z = eps * exp(lv/2) + mu
__global__ __launch_bounds__(BLK) void gauss_head_fwd_kernel(
    const float* __restrict__ hin, const float* __restrict__ Wm,
    const float* __restrict__ bm, const float* __restrict__ Wl,
    const float* __restrict__ bl, const float* __restrict__ eps,
    float* __restrict__ mu, float* __restrict__ lv, float* __restrict__ z,
    int B, int N, int K) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)B * N;
       i += (long)gridDim.x * BLK) {
    const int r = (int)(i / N);
    const int n = (int)(i - (long)r * N);
    const float* hr = hin + (long)r * K;
    const float* wm = Wm + (long)n * K;
    const float* wl = Wl + (long)n * K;
    float am = bm[n], al = bl[n];
    for (int k = 0; k < K; ++k) {
      const float hv = hr[k];
      am += wm[k] * hv;
      al += wl[k] * hv;
    }
    mu[i] = am;
    lv[i] = al;
    z[i] = eps[i] * __expf(0.5f * al) + am;
  }
}

// combine head grads: dmu_t = dmu + dz, dlv_t = dlv + dz*eps*exp(lv/2)/2
__global__ __launch_bounds__(BLK) void gauss_head_combine_kernel(
    const float* __restrict__ dz, const float* __restrict__ dmu,
    const float* __restrict__ dlv, const float* __restrict__ eps,
    const float* __restrict__ lv, float* __restrict__ dmu_t,
    float* __restrict__ dlv_t, long nel) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < nel;
       i += (long)gridDim.x * BLK) {
    const float dzv = dz != nullptr ? dz[i] : 0.f;
    dmu_t[i] = (dmu != nullptr ? dmu[i] : 0.f) + dzv;
    dlv_t[i] = (dlv != nullptr ? dlv[i] : 0.f) +
               dzv * eps[i] * 0.5f * __expf(0.5f * lv[i]);
  }
}

// dh[r][k] = dmu_t @ Wm + dlv_t @ Wl  (N is tiny: 10)
__global__ __launch_bounds__(BLK) void gauss_head_dh_kernel(
    const float* __restrict__ dmu_t, const float* __restrict__ dlv_t,
    const float* __restrict__ Wm, const float* __restrict__ Wl,
    float* __restrict__ dh, int B, int N, int K) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)B * K;
       i += (long)gridDim.x * BLK) {
    const int r = (int)(i / K);
    const int k = (int)(i - (long)r * K);
    float acc = 0.f;
    for (int n = 0; n < N; ++n) {
      acc += dmu_t[(long)r * N + n] * Wm[(long)n * K + k] +
             dlv_t[(long)r * N + n] * Wl[(long)n * K + k];
    }
    dh[i] = acc;
  }
}

// dW[n][k] (+)= gout^T @ hin, db[n] (+)= colsum(gout) — plain Linear grad
__global__ __launch_bounds__(BLK) void linear_dw_kernel(
    const float* __restrict__ gout, const float* __restrict__ hin,
    float* __restrict__ dW, float* __restrict__ db, int B, int N, int K,
    int accumulate) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)N * (K + 1);
       i += (long)gridDim.x * BLK) {
    const int n = (int)(i / (K + 1));
    const int k = (int)(i - (long)n * (K + 1));
    float acc = 0.f;
    if (k == K) {
      if (db == nullptr) continue;
      for (int r = 0; r < B; ++r) acc += gout[(long)r * N + n];
      db[n] = accumulate ? db[n] + acc : acc;
      continue;
    }
    for (int r = 0; r < B; ++r)
      acc += gout[(long)r * N + n] * hin[(long)r * K + k];
    dW[(long)n * K + k] = accumulate ? dW[(long)n * K + k] + acc : acc;
  }
}

// out = tanh(h @ W^T + b); saves the post-tanh value (bwd recomputes 1-y^2)
__global__ __launch_bounds__(BLK) void tanh_head_fwd_kernel(
    const float* __restrict__ hin, const float* __restrict__ W,
    const float* __restrict__ b, float* __restrict__ out, int B, int N,
    int K) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < (long)B * N;
       i += (long)gridDim.x * BLK) {
    const int r = (int)(i / N);
    const int n = (int)(i - (long)r * N);
    const float* hr = hin + (long)r * K;
    const float* w = W + (long)n * K;
    float acc = b[n];
    for (int k = 0; k < K; ++k) acc += w[k] * hr[k];
    out[i] = tanhf(acc);
  }
}

// dpre = dy * (1 - y^2), written out for the dW kernel; dh = dpre @ W
__global__ __launch_bounds__(BLK) void tanh_head_dpre_kernel(
    const float* __restrict__ dy, const float* __restrict__ y,
    float* __restrict__ dpre, long nel) {
  for (long i = (long)blockIdx.x * BLK + threadIdx.x; i < nel;
       i += (long)gridDim.x * BLK) {
    dpre[i] = dy[i] * (1.f - y[i] * y[i]);
  }
}

int pgrid(long n) { return (int)std::min<long>(2048, (n + BLK - 1) / BLK); }

void chk(const torch::Tensor& t) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              "lstm heads: fp32 contiguous CUDA tensors required");
}

}  // namespace

// out = [h | g | s1 | s2] @ W^T + b (concat-free embed projection)
torch::Tensor affine4_fwd(torch::Tensor h, torch::Tensor g, torch::Tensor s1,
                          torch::Tensor s2, torch::Tensor W,
                          c10::optional<torch::Tensor> b) {
  chk(h); chk(g); chk(W);
  const int B = h.size(0), K1 = h.size(1), K2 = g.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == K1 + K2 + 2, "affine4: W width mismatch");
  auto out = torch::empty({B, N}, h.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(affine4_fwd_kernel, dim3(pgrid((long)B * N)), dim3(BLK),
                     0, stream, h.data_ptr<float>(), g.data_ptr<float>(),
                     s1.data_ptr<float>(), s2.data_ptr<float>(),
                     W.data_ptr<float>(),
                     b.has_value() ? b->data_ptr<float>() : nullptr,
                     out.data_ptr<float>(), B, N, K1, K2);
  return out;
}

// returns (dh, dg); accumulates dW/db into the given buffers when present
std::vector<torch::Tensor> affine4_bwd(
    torch::Tensor gout, torch::Tensor h, torch::Tensor g, torch::Tensor s1,
    torch::Tensor s2, torch::Tensor W, c10::optional<torch::Tensor> dW_acc,
    c10::optional<torch::Tensor> db_acc, bool need_dh, bool need_dg) {
  const int B = h.size(0), K1 = h.size(1), K2 = g.size(1), N = W.size(0);
  const int K = K1 + K2 + 2;
  auto stream = at::cuda::getCurrentCUDAStream();
  torch::Tensor dh, dg;
  if (need_dh) {
    dh = torch::empty_like(h);
    hipLaunchKernelGGL(affine_dsrc_kernel, dim3(pgrid((long)B * K1)),
                       dim3(BLK), 0, stream, gout.data_ptr<float>(),
                       W.data_ptr<float>(), dh.data_ptr<float>(), B, N, K,
                       K1, 0);
  }
  if (need_dg) {
    dg = torch::empty_like(g);
    hipLaunchKernelGGL(affine_dsrc_kernel, dim3(pgrid((long)B * K2)),
                       dim3(BLK), 0, stream, gout.data_ptr<float>(),
                       W.data_ptr<float>(), dg.data_ptr<float>(), B, N, K,
                       K2, K1);
  }
  if (dW_acc.has_value()) {
    hipLaunchKernelGGL(affine4_dw_kernel, dim3(pgrid((long)N * (K + 1))),
                       dim3(BLK), 0, stream, gout.data_ptr<float>(),
                       h.data_ptr<float>(), g.data_ptr<float>(),
                       s1.data_ptr<float>(), s2.data_ptr<float>(),
                       dW_acc->data_ptr<float>(),
                       db_acc.has_value() ? db_acc->data_ptr<float>() : nullptr,
                       B, N, K1, K2, 1);
  }
  return {dh, dg};
}

std::vector<torch::Tensor> gauss_head_fwd(torch::Tensor hin, torch::Tensor Wm,
                                          torch::Tensor bm, torch::Tensor Wl,
                                          torch::Tensor bl,
                                          torch::Tensor eps) {
  chk(hin); chk(Wm); chk(Wl);
  const int B = hin.size(0), K = hin.size(1), N = Wm.size(0);
  auto mu = torch::empty({B, N}, hin.options());
  auto lv = torch::empty({B, N}, hin.options());
  auto z = torch::empty({B, N}, hin.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gauss_head_fwd_kernel, dim3(pgrid((long)B * N)),
                     dim3(BLK), 0, stream, hin.data_ptr<float>(),
                     Wm.data_ptr<float>(), bm.data_ptr<float>(),
                     Wl.data_ptr<float>(), bl.data_ptr<float>(),
                     eps.data_ptr<float>(), mu.data_ptr<float>(),
                     lv.data_ptr<float>(), z.data_ptr<float>(), B, N, K);
  return {mu, lv, z};
}

// returns dh; accumulates the four head-param grads when buffers present
torch::Tensor gauss_head_bwd(
    c10::optional<torch::Tensor> dz, c10::optional<torch::Tensor> dmu,
    c10::optional<torch::Tensor> dlv, torch::Tensor eps, torch::Tensor lv,
    torch::Tensor hin, torch::Tensor Wm, torch::Tensor Wl,
    c10::optional<torch::Tensor> dWm, c10::optional<torch::Tensor> dbm,
    c10::optional<torch::Tensor> dWl, c10::optional<torch::Tensor> dbl) {
  const int B = hin.size(0), K = hin.size(1), N = Wm.size(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto dmu_t = torch::empty({B, N}, hin.options());
  auto dlv_t = torch::empty({B, N}, hin.options());
  const long nel = (long)B * N;
  hipLaunchKernelGGL(gauss_head_combine_kernel, dim3(pgrid(nel)), dim3(BLK),
                     0, stream,
                     dz.has_value() ? dz->data_ptr<float>() : nullptr,
                     dmu.has_value() ? dmu->data_ptr<float>() : nullptr,
                     dlv.has_value() ? dlv->data_ptr<float>() : nullptr,
                     eps.data_ptr<float>(), lv.data_ptr<float>(),
                     dmu_t.data_ptr<float>(), dlv_t.data_ptr<float>(), nel);
  auto dh = torch::empty_like(hin);
  hipLaunchKernelGGL(gauss_head_dh_kernel, dim3(pgrid((long)B * K)),
                     dim3(BLK), 0, stream, dmu_t.data_ptr<float>(),
                     dlv_t.data_ptr<float>(), Wm.data_ptr<float>(),
                     Wl.data_ptr<float>(), dh.data_ptr<float>(), B, N, K);
  if (dWm.has_value()) {
    hipLaunchKernelGGL(linear_dw_kernel, dim3(pgrid((long)N * (K + 1))),
                       dim3(BLK), 0, stream, dmu_t.data_ptr<float>(),
                       hin.data_ptr<float>(), dWm->data_ptr<float>(),
                       dbm.has_value() ? dbm->data_ptr<float>() : nullptr,
                       B, N, K, 1);
  }
  if (dWl.has_value()) {
    hipLaunchKernelGGL(linear_dw_kernel, dim3(pgrid((long)N * (K + 1))),
                       dim3(BLK), 0, stream, dlv_t.data_ptr<float>(),
                       hin.data_ptr<float>(), dWl->data_ptr<float>(),
                       dbl.has_value() ? dbl->data_ptr<float>() : nullptr,
                       B, N, K, 1);
  }
  return dh;
}

torch::Tensor tanh_head_fwd(torch::Tensor hin, torch::Tensor W,
                            torch::Tensor b) {
  chk(hin); chk(W);
  const int B = hin.size(0), K = hin.size(1), N = W.size(0);
  auto out = torch::empty({B, N}, hin.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tanh_head_fwd_kernel, dim3(pgrid((long)B * N)),
                     dim3(BLK), 0, stream, hin.data_ptr<float>(),
                     W.data_ptr<float>(), b.data_ptr<float>(),
                     out.data_ptr<float>(), B, N, K);
  return out;
}

// returns dh; accumulates dW/db when buffers present
torch::Tensor tanh_head_bwd(torch::Tensor dy, torch::Tensor y,
                            torch::Tensor hin, torch::Tensor W,
                            c10::optional<torch::Tensor> dW_acc,
                            c10::optional<torch::Tensor> db_acc) {
  const int B = hin.size(0), K = hin.size(1), N = W.size(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto dpre = torch::empty_like(y);
  const long nel = (long)B * N;
  hipLaunchKernelGGL(tanh_head_dpre_kernel, dim3(pgrid(nel)), dim3(BLK), 0,
                     stream, dy.data_ptr<float>(), y.data_ptr<float>(),
                     dpre.data_ptr<float>(), nel);
  auto dh = torch::empty_like(hin);
  hipLaunchKernelGGL(affine_dsrc_kernel, dim3(pgrid((long)B * K)), dim3(BLK),
                     0, stream, dpre.data_ptr<float>(), W.data_ptr<float>(),
                     dh.data_ptr<float>(), B, N, K, K, 0);
  if (dW_acc.has_value()) {
    hipLaunchKernelGGL(linear_dw_kernel, dim3(pgrid((long)N * (K + 1))),
                       dim3(BLK), 0, stream, dpre.data_ptr<float>(),
                       hin.data_ptr<float>(), dW_acc->data_ptr<float>(),
                       db_acc.has_value() ? db_acc->data_ptr<float>() : nullptr,
                       B, N, K, 1);
  }
  return dh;
}
