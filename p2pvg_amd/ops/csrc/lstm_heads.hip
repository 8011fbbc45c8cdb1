// Fused LSTM stack heads for gfx950.
//
// The reference assembles every LSTM input with torch.cat and runs the
// embed / mu / logvar / output Linears as separate GEMMs
// (reference models/lstm.py:13,16,56-57, models/p2p_model.py:241-247).
// Here each head is ONE kernel each way, built on a shared LDS-tiled
// fp32 GEMM skeleton (the lstm_cell kernel's 64-col x 32-row x 64-k
// structure): W tiles are staged COALESCED and read by all rows of the
// block, so W traffic is B/32 x |W| instead of B x |W| (the first, naive
// thread-per-output version streamed W per row and lost to hipBLASLt).
//
// - head_fwd: out = [h | g] @ W^T (+ s1*w_ks + s2*w_ks1) + b with an
//   activation/epilogue mode: 0 plain (affine4 embed), 1 tanh (output
//   head), 2 gauss (stacked mu/lv rows + z = eps*exp(lv/2)+mu).
// - head_dsrc: d[h|g] = gout @ W with dual destinations (the concat-free
//   backward split).
// - head_dw: dW[n][k] (+)= gout^T @ [h | g | s1 | s2 | 1] — the bias lives
//   in the last virtual column; one thread owns one (n,k): deterministic,
//   accumulates straight into the managed fp32 .grad buffers.
// All fp32 (the recurrent path stays fp32 under autocast).

#include "common.h"

namespace {

constexpr int TC = 64;     // output columns per block
constexpr int RB = 32;     // batch rows per block
constexpr int KS = 64;     // k-chunk
constexpr int THREADS = 256;

struct __align__(16) HeadLds {
  float w[KS][TC + 1];
  float a[RB][KS + 1];
  float outs[RB][TC];      // gauss epilogue cross-column staging
};

// virtual dense source: k < K1 -> h, else g
__device__ __forceinline__ float vsrc(const float* __restrict__ h,
                                      const float* __restrict__ g, int r,
                                      int k, int K1, int K2) {
  if (k < K1) return h[(long)r * K1 + k];
  const int kk = k - K1;
  return kk < K2 ? g[(long)r * K2 + kk] : 0.f;
}

// out(B,N) = [h|g] @ W^T + b (+ scalar columns when s1 != null).
// mode 0: plain; 1: tanh; 2: gauss stacked (N = 2*Nz rows: mu then lv;
// writes mu, lv, z — requires N <= TC so both heads share the block).
__global__ __launch_bounds__(THREADS) void head_fwd_kernel(
    const float* __restrict__ h, const float* __restrict__ g,
    const float* __restrict__ s1, const float* __restrict__ s2,
    const float* __restrict__ W, const float* __restrict__ b,
    const float* __restrict__ eps, float* __restrict__ out,
    float* __restrict__ out_lv, float* __restrict__ out_z, int B, int N,
    int K1, int K2, int mode) {
  __shared__ HeadLds lds;
  const int Kd = K1 + K2;
  const int Kw = g == nullptr ? K1 : (s1 != nullptr ? Kd + 2 : Kd);
  const int n0 = blockIdx.x * TC;
  const int r0 = blockIdx.y * RB;
  const int tid = threadIdx.x;
  const int c_idx = tid & (TC - 1);
  const int r_par = tid >> 6;
  constexpr int RT = RB / 4;
  float acc[RT];
#pragma unroll
  for (int t = 0; t < RT; ++t) acc[t] = 0.f;
  const int rows = min(RB, B - r0);

  for (int k0 = 0; k0 < Kd; k0 += KS) {
    {
      const int cc = tid >> 2;
      const int kk0 = (tid & 3) * 16;
      const float* wrow =
          n0 + cc < N ? W + (long)(n0 + cc) * Kw + k0 : nullptr;
#pragma unroll
      for (int kk = 0; kk < 16; ++kk) {
        const int k = k0 + kk0 + kk;
        lds.w[kk0 + kk][cc] =
            (wrow != nullptr && k < Kd) ? wrow[kk0 + kk] : 0.f;
      }
    }
    {
      const int r = tid >> 3;
      const int kk0 = (tid & 7) * 8;
      if (r < rows) {
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          const int k = k0 + kk0 + kk;
          lds.a[r][kk0 + kk] = k < Kd ? vsrc(h, g, r0 + r, k, K1, K2) : 0.f;
        }
      }
    }
    __syncthreads();
    const int klim = min(KS, Kd - k0);
    for (int kk = 0; kk < klim; ++kk) {
      const float wv = lds.w[kk][c_idx];
#pragma unroll
      for (int t = 0; t < RT; ++t)
        acc[t] = fmaf(lds.a[r_par + 4 * t][kk], wv, acc[t]);
    }
    __syncthreads();
  }

  const int n = n0 + c_idx;
  const bool colv = n < N;
  const float bias = (colv && b != nullptr) ? b[n] : 0.f;
  const float ws1 = (colv && s1 != nullptr) ? W[(long)n * Kw + Kd] : 0.f;
  const float ws2 = (colv && s1 != nullptr) ? W[(long)n * Kw + Kd + 1] : 0.f;
#pragma unroll
  for (int t = 0; t < RT; ++t) {
    const int r = r_par + 4 * t;
    if (r >= rows || !colv) continue;
    float v = acc[t] + bias;
    if (s1 != nullptr)
      v += s1[r0 + r] * ws1 + s2[r0 + r] * ws2;
    if (mode == 1) v = tanhf(v);
    if (mode == 2) {
      lds.outs[r][c_idx] = v;
    } else {
      out[(long)(r0 + r) * N + n] = v;
    }
  }
  if (mode == 2) {
    __syncthreads();
    // stacked gauss: rows [0,Nz) = mu, [Nz,2Nz) = lv; z from both
    const int Nz = N / 2;
    for (int i = tid; i < rows * Nz; i += THREADS) {
      const int r = i / Nz;
      const int j = i - r * Nz;
      const float mu = lds.outs[r][j];
      const float lv = lds.outs[r][Nz + j];
      const long off = (long)(r0 + r) * Nz + j;
      out[off] = mu;
      out_lv[off] = lv;
      out_z[off] = eps[off] * __expf(0.5f * lv) + mu;
    }
  }
}

// d[h|g](B, K1+K2) = gout(B,N) @ W(N, Kw)[:, :K1+K2] — dual destinations
__global__ __launch_bounds__(THREADS) void head_dsrc_kernel(
    const float* __restrict__ gout, const float* __restrict__ W,
    float* __restrict__ dh, float* __restrict__ dg, int B, int N, int K1,
    int K2, int Kw) {
  __shared__ HeadLds lds;
  const int Kd = K1 + K2;
  const int k0 = blockIdx.x * TC;       // output K columns
  const int r0 = blockIdx.y * RB;
  const int tid = threadIdx.x;
  const int c_idx = tid & (TC - 1);
  const int r_par = tid >> 6;
  constexpr int RT = RB / 4;
  float acc[RT];
#pragma unroll
  for (int t = 0; t < RT; ++t) acc[t] = 0.f;
  const int rows = min(RB, B - r0);

  for (int nc0 = 0; nc0 < N; nc0 += KS) {
    {
      // stage W[nc0+nn][k0+cc] -> w[nn][cc]: coalesced in k
      const int nn = tid >> 2;
      const int cc0 = (tid & 3) * 16;
      const float* wrow =
          nc0 + nn < N ? W + (long)(nc0 + nn) * Kw + k0 : nullptr;
#pragma unroll
      for (int cc = 0; cc < 16; ++cc) {
        const int k = k0 + cc0 + cc;
        lds.w[nn][cc0 + cc] =
            (wrow != nullptr && k < Kd) ? wrow[cc0 + cc] : 0.f;
      }
    }
    {
      const int r = tid >> 3;
      const int nn0 = (tid & 7) * 8;
      if (r < rows) {
        const float* gr = gout + (long)(r0 + r) * N + nc0;
#pragma unroll
        for (int nn = 0; nn < 8; ++nn) {
          lds.a[r][nn0 + nn] = nc0 + nn0 + nn < N ? gr[nn0 + nn] : 0.f;
        }
      }
    }
    __syncthreads();
    const int nlim = min(KS, N - nc0);
    for (int nn = 0; nn < nlim; ++nn) {
      const float wv = lds.w[nn][c_idx];
#pragma unroll
      for (int t = 0; t < RT; ++t)
        acc[t] = fmaf(lds.a[r_par + 4 * t][nn], wv, acc[t]);
    }
    __syncthreads();
  }

  const int k = k0 + c_idx;
#pragma unroll
  for (int t = 0; t < RT; ++t) {
    const int r = r_par + 4 * t;
    if (r >= rows || k >= Kd) continue;
    if (k < K1) {
      if (dh != nullptr) dh[(long)(r0 + r) * K1 + k] = acc[t];
    } else if (dg != nullptr) {
      dg[(long)(r0 + r) * K2 + (k - K1)] = acc[t];
    }
  }
}

// dW(N, Kw) (+)= gout(B,N)^T @ [h | g | s1 | s2], db (+)= colsum(gout).
// Block tile: 64 n x 64 virtual-k, r-chunks of RB staged in LDS; each
// thread owns 16 (n,k) cells — deterministic, no atomics.
__global__ __launch_bounds__(THREADS) void head_dw_kernel(
    const float* __restrict__ gout, const float* __restrict__ h,
    const float* __restrict__ g, const float* __restrict__ s1,
    const float* __restrict__ s2, float* __restrict__ dW,
    float* __restrict__ db, int B, int N, int ldg, int K1, int K2, int Kw,
    int accumulate) {
  __shared__ struct __align__(16) {
    float gt[RB][TC + 1];   // gout tile (r, n)
    float at[RB][TC + 1];   // source tile (r, k)
  } lds;
  const int Kd = K1 + K2;
  const int Kv = Kw + 1;                 // + bias column
  const int n0 = blockIdx.x * TC;
  const int k0 = blockIdx.y * TC;
  const int tid = threadIdx.x;
  // thread grid 16x16 over the 64x64 tile: 4x4 cells per thread
  const int tn = (tid & 15) * 4;
  const int tk = (tid >> 4) * 4;
  float acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = 0.f;

  for (int r0 = 0; r0 < B; r0 += RB) {
    const int rows = min(RB, B - r0);
    {
      const int r = tid >> 3;
      const int c0 = (tid & 7) * 8;
      if (r < rows) {
        const float* gr = gout + (long)(r0 + r) * ldg + n0;
#pragma unroll
        for (int c = 0; c < 8; ++c)
          lds.gt[r][c0 + c] = n0 + c0 + c < N ? gr[c0 + c] : 0.f;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          const int k = k0 + c0 + c;
          float v = 0.f;
          if (k < Kd) v = vsrc(h, g, r0 + r, k, K1, K2);
          else if (k == Kd && s1 != nullptr) v = s1[r0 + r];
          else if (k == Kd + 1 && s1 != nullptr) v = s2[r0 + r];
          else if (k == Kv - 1) v = 1.f;           // bias column
          lds.at[r][c0 + c] = v;
        }
      } else {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          lds.gt[r][c0 + c] = 0.f;
          lds.at[r][c0 + c] = 0.f;
        }
      }
    }
    __syncthreads();
    for (int r = 0; r < RB; ++r) {
      float gv[4], av[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) gv[i] = lds.gt[r][tn + i];
#pragma unroll
      for (int j = 0; j < 4; ++j) av[j] = lds.at[r][tk + j];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = fmaf(gv[i], av[j], acc[i][j]);
    }
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int n = n0 + tn + i;
    if (n >= N) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int k = k0 + tk + j;
      if (k >= Kv) continue;
      if (k == Kv - 1) {
        if (db != nullptr)
          db[n] = accumulate ? db[n] + acc[i][j] : acc[i][j];
      } else if (k < Kw) {
        float* d = dW + (long)n * Kw + k;
        *d = accumulate ? *d + acc[i][j] : acc[i][j];
      }
    }
  }
}

// pointwise: dmu_t = dmu + dz, dlv_t = dlv + dz*eps*exp(lv/2)/2
// writes the two halves of the STACKED (B, 2N) grad directly (row stride
// 2N): no temporaries, no narrow-copy passes
__global__ __launch_bounds__(THREADS) void gauss_combine_kernel(
    const float* __restrict__ dz, const float* __restrict__ dmu,
    const float* __restrict__ dlv, const float* __restrict__ eps,
    const float* __restrict__ lv, float* __restrict__ dst, int N,
    long nel) {
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < nel;
       i += (long)gridDim.x * THREADS) {
    const long r = i / N;
    const int n = (int)(i - r * N);
    const float dzv = dz != nullptr ? dz[i] : 0.f;
    dst[r * 2 * N + n] = (dmu != nullptr ? dmu[i] : 0.f) + dzv;
    dst[r * 2 * N + N + n] =
        (dlv != nullptr ? dlv[i] : 0.f) +
        dzv * eps[i] * 0.5f * __expf(0.5f * lv[i]);
  }
}

// pointwise: dpre = dy * (1 - y^2)
__global__ __launch_bounds__(THREADS) void tanh_dpre_kernel(
    const float* __restrict__ dy, const float* __restrict__ y,
    float* __restrict__ dpre, long nel) {
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < nel;
       i += (long)gridDim.x * THREADS) {
    dpre[i] = dy[i] * (1.f - y[i] * y[i]);
  }
}

void chk(const torch::Tensor& t) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              "lstm heads: fp32 contiguous CUDA tensors required");
}

dim3 fwd_grid(int B, int N) {
  return dim3(ceil_div(N, TC), ceil_div(B, RB));
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
  hipLaunchKernelGGL(head_fwd_kernel, fwd_grid(B, N), dim3(THREADS), 0,
                     stream, h.data_ptr<float>(), g.data_ptr<float>(),
                     s1.data_ptr<float>(), s2.data_ptr<float>(),
                     W.data_ptr<float>(),
                     b.has_value() ? b->data_ptr<float>() : nullptr, nullptr,
                     out.data_ptr<float>(), nullptr, nullptr, B, N, K1, K2,
                     0);
  return out;
}

// returns (dh, dg); accumulates dW/db into the given buffers when present
std::vector<torch::Tensor> affine4_bwd(
    torch::Tensor gout, torch::Tensor h, torch::Tensor g, torch::Tensor s1,
    torch::Tensor s2, torch::Tensor W, c10::optional<torch::Tensor> dW_acc,
    c10::optional<torch::Tensor> db_acc, bool need_dh, bool need_dg) {
  const int B = h.size(0), K1 = h.size(1), K2 = g.size(1), N = W.size(0);
  const int Kw = K1 + K2 + 2;
  auto stream = at::cuda::getCurrentCUDAStream();
  torch::Tensor dh, dg;
  if (need_dh || need_dg) {
    if (need_dh) dh = torch::empty_like(h);
    if (need_dg) dg = torch::empty_like(g);
    hipLaunchKernelGGL(head_dsrc_kernel,
                       dim3(ceil_div(K1 + K2, TC), ceil_div(B, RB)),
                       dim3(THREADS), 0, stream, gout.data_ptr<float>(),
                       W.data_ptr<float>(),
                       need_dh ? dh.data_ptr<float>() : nullptr,
                       need_dg ? dg.data_ptr<float>() : nullptr, B, N, K1,
                       K2, Kw);
  }
  if (dW_acc.has_value()) {
    hipLaunchKernelGGL(head_dw_kernel,
                       dim3(ceil_div(N, TC), ceil_div(Kw + 1, TC)),
                       dim3(THREADS), 0, stream, gout.data_ptr<float>(),
                       h.data_ptr<float>(), g.data_ptr<float>(),
                       s1.data_ptr<float>(), s2.data_ptr<float>(),
                       dW_acc->data_ptr<float>(),
                       db_acc.has_value() ? db_acc->data_ptr<float>() : nullptr,
                       B, N, N, K1, K2, Kw, 1);
  }
  return {dh, dg};
}

std::vector<torch::Tensor> gauss_head_fwd(torch::Tensor hin, torch::Tensor Wm,
                                          torch::Tensor bm, torch::Tensor Wl,
                                          torch::Tensor bl, torch::Tensor eps,
                                          c10::optional<torch::Tensor> Ws_c,
                                          c10::optional<torch::Tensor> bs_c) {
  chk(hin); chk(Wm); chk(Wl);
  const int B = hin.size(0), K = hin.size(1), N = Wm.size(0);
  TORCH_CHECK(2 * N <= TC, "gauss head: 2*z_dim must fit one column tile");
  // stacked weights/biases: rows [0,N) = mu head, [N,2N) = lv head —
  // either the caller's per-step cache or a fresh cat
  auto Ws = Ws_c.has_value() ? Ws_c.value()
                             : torch::cat({Wm, Wl}, 0).contiguous();
  auto bs = bs_c.has_value() ? bs_c.value()
                             : torch::cat({bm, bl}, 0).contiguous();
  auto mu = torch::empty({B, N}, hin.options());
  auto lv = torch::empty({B, N}, hin.options());
  auto z = torch::empty({B, N}, hin.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(head_fwd_kernel, fwd_grid(B, 2 * N), dim3(THREADS), 0,
                     stream, hin.data_ptr<float>(), nullptr, nullptr,
                     nullptr, Ws.data_ptr<float>(), bs.data_ptr<float>(),
                     eps.data_ptr<float>(), mu.data_ptr<float>(),
                     lv.data_ptr<float>(), z.data_ptr<float>(), B, 2 * N, K,
                     0, 2);
  return {mu, lv, z, Ws};
}

// returns dh; accumulates the four head-param grads when buffers present
torch::Tensor gauss_head_bwd(
    c10::optional<torch::Tensor> dz, c10::optional<torch::Tensor> dmu,
    c10::optional<torch::Tensor> dlv, torch::Tensor eps, torch::Tensor lv,
    torch::Tensor hin, torch::Tensor Ws, long N_,
    c10::optional<torch::Tensor> dWm, c10::optional<torch::Tensor> dbm,
    c10::optional<torch::Tensor> dWl, c10::optional<torch::Tensor> dbl) {
  const int B = hin.size(0), K = hin.size(1), N = (int)N_;
  auto stream = at::cuda::getCurrentCUDAStream();
  // stacked head-grad (B, 2N): [dmu_t | dlv_t], written strided in one pass
  auto dst = torch::empty({B, 2 * N}, hin.options());
  const long nel = (long)B * N;
  hipLaunchKernelGGL(gauss_combine_kernel,
                     dim3((int)std::min<long>(2048, (nel + THREADS - 1) / THREADS)),
                     dim3(THREADS), 0, stream,
                     dz.has_value() ? dz->data_ptr<float>() : nullptr,
                     dmu.has_value() ? dmu->data_ptr<float>() : nullptr,
                     dlv.has_value() ? dlv->data_ptr<float>() : nullptr,
                     eps.data_ptr<float>(), lv.data_ptr<float>(),
                     dst.data_ptr<float>(), N, nel);
  auto dh = torch::empty_like(hin);
  hipLaunchKernelGGL(head_dsrc_kernel, dim3(ceil_div(K, TC), ceil_div(B, RB)),
                     dim3(THREADS), 0, stream, dst.data_ptr<float>(),
                     Ws.data_ptr<float>(), dh.data_ptr<float>(), nullptr, B,
                     2 * N, K, 0, K);
  // the two dW kernels read their half of the stacked grad IN PLACE
  // (row stride 2N) — no contiguous() copies
  if (dWm.has_value()) {
    hipLaunchKernelGGL(head_dw_kernel,
                       dim3(ceil_div(N, TC), ceil_div(K + 1, TC)),
                       dim3(THREADS), 0, stream, dst.data_ptr<float>(),
                       hin.data_ptr<float>(), nullptr, nullptr, nullptr,
                       dWm->data_ptr<float>(),
                       dbm.has_value() ? dbm->data_ptr<float>() : nullptr,
                       B, N, 2 * N, K, 0, K, 1);
  }
  if (dWl.has_value()) {
    hipLaunchKernelGGL(head_dw_kernel,
                       dim3(ceil_div(N, TC), ceil_div(K + 1, TC)),
                       dim3(THREADS), 0, stream, dst.data_ptr<float>() + N,
                       hin.data_ptr<float>(), nullptr, nullptr, nullptr,
                       dWl->data_ptr<float>(),
                       dbl.has_value() ? dbl->data_ptr<float>() : nullptr,
                       B, N, 2 * N, K, 0, K, 1);
  }
  return dh;
}

torch::Tensor tanh_head_fwd(torch::Tensor hin, torch::Tensor W,
                            torch::Tensor b) {
  chk(hin); chk(W);
  const int B = hin.size(0), K = hin.size(1), N = W.size(0);
  auto out = torch::empty({B, N}, hin.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(head_fwd_kernel, fwd_grid(B, N), dim3(THREADS), 0,
                     stream, hin.data_ptr<float>(), nullptr, nullptr,
                     nullptr, W.data_ptr<float>(), b.data_ptr<float>(),
                     nullptr, out.data_ptr<float>(), nullptr, nullptr, B, N,
                     K, 0, 1);
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
  hipLaunchKernelGGL(tanh_dpre_kernel,
                     dim3((int)std::min<long>(2048, (nel + THREADS - 1) / THREADS)),
                     dim3(THREADS), 0, stream, dy.data_ptr<float>(),
                     y.data_ptr<float>(), dpre.data_ptr<float>(), nel);
  auto dh = torch::empty_like(hin);
  hipLaunchKernelGGL(head_dsrc_kernel, dim3(ceil_div(K, TC), ceil_div(B, RB)),
                     dim3(THREADS), 0, stream, dpre.data_ptr<float>(),
                     W.data_ptr<float>(), dh.data_ptr<float>(), nullptr, B,
                     N, K, 0, K);
  if (dW_acc.has_value()) {
    hipLaunchKernelGGL(head_dw_kernel,
                       dim3(ceil_div(N, TC), ceil_div(K + 1, TC)),
                       dim3(THREADS), 0, stream, dpre.data_ptr<float>(),
                       hin.data_ptr<float>(), nullptr, nullptr, nullptr,
                       dW_acc->data_ptr<float>(),
                       db_acc.has_value() ? db_acc->data_ptr<float>() : nullptr,
                       B, N, N, K, 0, K, 1);
  }
  return dh;
}
