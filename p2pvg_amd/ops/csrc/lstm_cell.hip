// Fused LSTM cell for gfx950.
//
// The reference launches nn.LSTMCell per stack per timestep
// (reference models/lstm.py:14,55), which on ROCm decomposes into ~6 ATen
// kernels (two GEMMs + bias + pointwise). Here the whole cell is ONE kernel:
// both GEMMs (x@W_ih^T + h@W_hh^T + biases), the four gate activations, and
// the state update, with activated gates saved for the backward.
//
// Shapes: x,h,c (B,H) fp32; W_ih,W_hh (4H,H) fp32; H=rnn_size (256 default),
// B = batch (4..192). This is a latency-bound op (the weights are ~1 MB,
// L2-resident); the win is one launch instead of six and LDS-staged operand
// reuse across the batch.
//
// Tiling: each workgroup owns J=16 hidden indices for all 4 gates
// (64 gate-columns: n = gate*H + j, so the epilogue has all four gates of a
// (b, j) pair inside the workgroup) and RB=32 batch rows. K-loop stages
// W tiles (64 cols x 64 k) and x/h tiles (32 x 64) in LDS, transposed so the
// inner product reads are conflict-free (w_lds[k][c]: lanes read consecutive
// c; 65-padding kills the power-of-2 stride).

#include "common.h"

namespace {

constexpr int TJ = 16;    // hidden indices per workgroup
constexpr int TC = 64;    // gate-columns per workgroup = 4*TJ
constexpr int RB = 32;    // batch rows per workgroup
constexpr int KS = 64;    // K-chunk
constexpr int THREADS = 256;

// LDS: w_i[k][c], w_h[k][c] (KS x TC, padded), xh[r][k] (RB x KS, padded x2)
struct __align__(16) LstmLds {
  float wi[KS][TC + 1];
  float wh[KS][TC + 1];
  float xs[RB][KS + 1];
  float hs[RB][KS + 1];
  float gates[RB][TC];   // activated gates staging for the epilogue transpose
};

__global__ __launch_bounds__(THREADS) void lstm_cell_fwd_kernel(
    const float* __restrict__ x,     // (B,H)
    const float* __restrict__ h,     // (B,H)
    const float* __restrict__ c,     // (B,H)
    const float* __restrict__ w_ih,  // (4H,H) [i,f,g,o] blocks
    const float* __restrict__ w_hh,  // (4H,H)
    const float* __restrict__ b_ih,  // (4H)
    const float* __restrict__ b_hh,  // (4H)
    float* __restrict__ h_out,       // (B,H)
    float* __restrict__ c_out,       // (B,H)
    float* __restrict__ gates_out,   // (B,4H) activated i,f,g,o (for bwd)
    int B, int H) {
  __shared__ LstmLds lds;

  const int j0 = blockIdx.x * TJ;   // hidden-index base
  const int r0 = blockIdx.y * RB;   // batch-row base
  const int tid = threadIdx.x;

  const int c_idx = tid & (TC - 1);       // 0..63: column within tile
  const int r_par = tid >> 6;             // 0..3: row partition
  const int gate = c_idx >> 4;            // 0..3 (i,f,g,o)
  const int j = j0 + (c_idx & (TJ - 1));  // hidden index
  const int n = gate * H + j;             // gate-column in (4H)

  // per-thread accumulators: rows r0 + r_par + 4*t
  constexpr int RT = RB / 4;  // 8 rows per thread
  float acc[RT];
#pragma unroll
  for (int t = 0; t < RT; ++t) acc[t] = 0.f;

  const int rows = min(RB, B - r0);

  for (int k0 = 0; k0 < H; k0 += KS) {
    // stage W tiles transposed: thread loads W[n0+cc][k0+kk] -> w[kk][cc]
    // 256 threads load 64x64 = 4096 elements: each thread 16, coalesced in k.
    {
      const int cc = tid >> 2;            // 0..63 column
      const int kk0 = (tid & 3) * 16;     // 16 k's per thread
      const int gcol = (cc >> 4) * H + j0 + (cc & 15);
      const float* wi_row = w_ih + (long)gcol * H + k0;
      const float* wh_row = w_hh + (long)gcol * H + k0;
#pragma unroll
      for (int kk = 0; kk < 16; ++kk) {
        lds.wi[kk0 + kk][cc] = wi_row[kk0 + kk];
        lds.wh[kk0 + kk][cc] = wh_row[kk0 + kk];
      }
    }
    // stage x/h tiles: RB x KS, coalesced in k
    {
      const int r = tid >> 3;             // 0..31
      const int kk0 = (tid & 7) * 8;      // 8 k's per thread
      if (r < rows) {
        const float* xr = x + (long)(r0 + r) * H + k0;
        const float* hr = h + (long)(r0 + r) * H + k0;
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          lds.xs[r][kk0 + kk] = xr[kk0 + kk];
          lds.hs[r][kk0 + kk] = hr[kk0 + kk];
        }
      }
    }
    __syncthreads();

#pragma unroll 4
    for (int kk = 0; kk < KS; ++kk) {
      const float wi = lds.wi[kk][c_idx];
      const float wh = lds.wh[kk][c_idx];
#pragma unroll
      for (int t = 0; t < RT; ++t) {
        const int r = r_par + 4 * t;
        acc[t] = fmaf(lds.xs[r][kk], wi, fmaf(lds.hs[r][kk], wh, acc[t]));
      }
    }
    __syncthreads();
  }

  // bias + activation, stash into LDS for the (gate, j) transpose
  const float bias = b_ih[n] + b_hh[n];
#pragma unroll
  for (int t = 0; t < RT; ++t) {
    const int r = r_par + 4 * t;
    if (r0 + r < B) {
      float v = acc[t] + bias;
      v = (gate == 2) ? tanhf(v) : sigmoidf_(v);
      lds.gates[r][c_idx] = v;
      gates_out[(long)(r0 + r) * 4 * H + n] = v;
    }
  }
  __syncthreads();

  // epilogue: threads (r, j) compute the state update
  // 256 threads over rows x TJ: tid -> jj = tid % TJ, r = tid / TJ (16 rows/pass)
  const int jj = tid & (TJ - 1);
  for (int r = tid >> 4; r < rows; r += THREADS / TJ) {
    const float gi = lds.gates[r][0 * TJ + jj];
    const float gf = lds.gates[r][1 * TJ + jj];
    const float gg = lds.gates[r][2 * TJ + jj];
    const float go = lds.gates[r][3 * TJ + jj];
    const long off = (long)(r0 + r) * H + j0 + jj;
    const float cn = gf * c[off] + gi * gg;
    c_out[off] = cn;
    h_out[off] = go * tanhf(cn);
  }
}

// Backward pointwise: from (dh, dc_in, gates, c_prev, c_new) produce
// pre-activation gate grads (B,4H) and dc_prev (B,H). The four GEMMs
// (dx, dh_prev, dW_ih, dW_hh) are plain GEMMs done with hipBLASLt via
// torch.mm in the autograd wrapper.
__global__ void lstm_cell_bwd_pointwise_kernel(
    const float* __restrict__ dh,        // (B,H)
    const float* __restrict__ dc_in,     // (B,H) may be null
    const float* __restrict__ gates,     // (B,4H) activated
    const float* __restrict__ c_prev,    // (B,H)
    const float* __restrict__ c_new,     // (B,H)
    float* __restrict__ dgates,          // (B,4H) pre-activation grads
    float* __restrict__ dc_prev,         // (B,H)
    int B, int H) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * H;
  if (idx >= total) return;
  const long b = idx / H;
  const long j = idx % H;
  const long g0 = b * 4 * H;

  const float gi = gates[g0 + 0 * H + j];
  const float gf = gates[g0 + 1 * H + j];
  const float gg = gates[g0 + 2 * H + j];
  const float go = gates[g0 + 3 * H + j];

  const float tcn = tanhf(c_new[idx]);
  const float dhv = dh[idx];
  float dc = dhv * go * (1.f - tcn * tcn);
  if (dc_in != nullptr) dc += dc_in[idx];

  const float di = dc * gg;
  const float df = dc * c_prev[idx];
  const float dg = dc * gi;
  const float do_ = dhv * tcn;

  dgates[g0 + 0 * H + j] = di * gi * (1.f - gi);
  dgates[g0 + 1 * H + j] = df * gf * (1.f - gf);
  dgates[g0 + 2 * H + j] = dg * (1.f - gg * gg);
  dgates[g0 + 3 * H + j] = do_ * go * (1.f - go);
  dc_prev[idx] = dc * gf;
}

}  // namespace

std::vector<torch::Tensor> lstm_cell_fwd(torch::Tensor x, torch::Tensor h,
                                         torch::Tensor c, torch::Tensor w_ih,
                                         torch::Tensor w_hh, torch::Tensor b_ih,
                                         torch::Tensor b_hh) {
  CHECK_INPUT(x);
  CHECK_INPUT(h);
  CHECK_INPUT(c);
  CHECK_INPUT(w_ih);
  CHECK_INPUT(w_hh);
  CHECK_INPUT(b_ih);
  CHECK_INPUT(b_hh);
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "lstm_cell_fwd: fp32 only");
  const int B = x.size(0);
  const int H = h.size(1);
  TORCH_CHECK(x.size(1) == H, "lstm_cell_fwd expects input_size == hidden_size");
  TORCH_CHECK(H % TJ == 0, "H must be a multiple of 16");

  auto h_out = torch::empty_like(h);
  auto c_out = torch::empty_like(c);
  auto gates = torch::empty({B, 4 * H}, x.options());

  dim3 grid(H / TJ, ceil_div(B, RB));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lstm_cell_fwd_kernel, grid, dim3(THREADS), 0, stream,
                     x.data_ptr<float>(), h.data_ptr<float>(), c.data_ptr<float>(),
                     w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
                     b_ih.data_ptr<float>(), b_hh.data_ptr<float>(),
                     h_out.data_ptr<float>(), c_out.data_ptr<float>(),
                     gates.data_ptr<float>(), B, H);
  return {h_out, c_out, gates};
}

std::vector<torch::Tensor> lstm_cell_bwd_pointwise(
    torch::Tensor dh, c10::optional<torch::Tensor> dc_in, torch::Tensor gates,
    torch::Tensor c_prev, torch::Tensor c_new) {
  CHECK_INPUT(dh);
  CHECK_INPUT(gates);
  CHECK_INPUT(c_prev);
  CHECK_INPUT(c_new);
  const int B = dh.size(0);
  const int H = dh.size(1);
  auto dgates = torch::empty({B, 4 * H}, dh.options());
  auto dc_prev = torch::empty_like(dh);
  const long total = (long)B * H;
  const int threads = 256;
  const int blocks = (int)((total + threads - 1) / threads);
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* dc_ptr = nullptr;
  if (dc_in.has_value()) {
    CHECK_INPUT(dc_in.value());
    dc_ptr = dc_in->data_ptr<float>();
  }
  hipLaunchKernelGGL(lstm_cell_bwd_pointwise_kernel, dim3(blocks), dim3(threads),
                     0, stream, dh.data_ptr<float>(), dc_ptr,
                     gates.data_ptr<float>(), c_prev.data_ptr<float>(),
                     c_new.data_ptr<float>(), dgates.data_ptr<float>(),
                     dc_prev.data_ptr<float>(), B, H);
  return {dgates, dc_prev};
}
