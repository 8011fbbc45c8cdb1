// Fused BatchNorm + activation for gfx950.
//
// The reference runs Conv2d -> BatchNorm2d -> LeakyReLU as three modules
// (reference models/dcgan_64.py:7-11), which on ROCm is MIOpen BN (two-pass)
// plus separate activation kernels each way. Here:
// - the conv epilogue already accumulates per-channel sum/sumsq (conv2d_nhwc
//   stats hook), so the forward needs only: finalize (K-sized kernel turning
//   sums into mean/invstd/scale/shift + running-stat update) and ONE
//   elementwise pass y = act(x*scale + shift).
// - backward folds the activation grad in: one reduction pass (s1 = sum dy',
//   s2 = sum dy'*xhat) and one apply pass for dx; dgamma/dbeta fall out of
//   s1/s2. fp32 statistics throughout (bf16-safe BN on tiny spatial maps is
//   SURVEY §7's "hard part" — stats never touch bf16).
//
// All tensors channels_last bf16; stats/params fp32.

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

namespace {

constexpr int BLOCK = 256;

__device__ __forceinline__ float act_fwd(float v, int act) {
  switch (act) {
    case 1: return v > 0.f ? v : 0.2f * v;
    case 2: return tanhf(v);
    case 3: return 1.f / (1.f + __expf(-v));
    default: return v;
  }
}

// derivative of act given the POST-activation value y
__device__ __forceinline__ float act_bwd_from_y(float y, int act) {
  switch (act) {
    case 1: return y > 0.f ? 1.f : 0.2f;
    case 2: return 1.f - y * y;
    case 3: return y * (1.f - y);
    default: return 1.f;
  }
}

// derivative of act given the PRE-activation value u (recomputing the cheap
// transcendental beats re-reading the y tensor from HBM: the backward kernels
// then touch only x and dy)
__device__ __forceinline__ float act_bwd_from_u(float u, int act) {
  switch (act) {
    case 1: return u > 0.f ? 1.f : 0.2f;
    case 2: {
      const float t = tanhf(u);
      return 1.f - t * t;
    }
    case 3: {
      const float s = 1.f / (1.f + __expf(-u));
      return s * (1.f - s);
    }
    default: return 1.f;
  }
}

// stats (2,K) raw sums -> saved mean/invstd + scale/shift (+ running update)
__global__ void bn_finalize_kernel(const float* __restrict__ stats,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ mean_out,
                                   float* __restrict__ invstd_out,
                                   float* __restrict__ scale_out,
                                   float* __restrict__ shift_out,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float momentum, float eps, float count,
                                   int K, int nbuckets) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= K) return;
  float s0 = 0.f, s1 = 0.f;
  for (int b = 0; b < nbuckets; ++b) {
    s0 += stats[(long)b * 2 * K + c];
    s1 += stats[(long)b * 2 * K + K + c];
  }
  const float m = s0 / count;
  float var = s1 / count - m * m;
  var = var > 0.f ? var : 0.f;
  const float inv = rsqrtf(var + eps);
  mean_out[c] = m;
  invstd_out[c] = inv;
  const float g = gamma[c], b = beta[c];
  scale_out[c] = g * inv;
  shift_out[c] = b - m * g * inv;
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    const float ub = var * (count / fmaxf(count - 1.f, 1.f));
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * ub;
  }
}

// eval mode: scale/shift straight from running stats
__global__ void bn_eval_prep_kernel(const float* __restrict__ running_mean,
                                    const float* __restrict__ running_var,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    float* __restrict__ mean_out,
                                    float* __restrict__ invstd_out,
                                    float* __restrict__ scale_out,
                                    float* __restrict__ shift_out, float eps,
                                    int K) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= K) return;
  const float m = running_mean[c];
  const float inv = rsqrtf(running_var[c] + eps);
  mean_out[c] = m;
  invstd_out[c] = inv;
  scale_out[c] = gamma[c] * inv;
  shift_out[c] = beta[c] - m * gamma[c] * inv;
}

// y = act(x*scale[c] + shift[c]); vectors of 8 bf16 along C.
// C/8 divides BLOCK and the grid stride, so each thread's channel group is
// fixed: per-channel scalars are preloaded into registers once.
// ring > 0: x and y are (N, C, H+2r, W+2r) padded maps whose logical content
// is the interior — the walk covers the FULL padded tensor, writing ZERO at
// ring positions (the next conv gathers the ring as its padding) and
// ignoring x's ring (which holds garbage from interior-only conv writes).
__global__ __launch_bounds__(BLOCK) void bn_act_fwd_kernel(
    const __bf16* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, __bf16* __restrict__ y, long nvec, int C,
    int act, int Hp, int Wp, int ring) {
  const int cvec = C / 8;
  const int c0 = (int)(((long)blockIdx.x * BLOCK + threadIdx.x) % cvec) * 8;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[c0 + j];
    sh[j] = shift[c0 + j];
  }
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float vv = (float)v[j] * sc[j] + sh[j];
      o[j] = (__bf16)act_fwd(vv, act);
    }
    *reinterpret_cast<bf16x8*>(y + i * 8) = o;
  }
}

// ------ ring > 0 row-walk variants ------
// One padded row (n, yy) per grid-stride block step; within a row the only
// index math is shifts/masks (cvec = C/8 is a power of two by the C/8 |
// 256 dispatch gate). Ring positions are written as zeros (fwd y / bwd dx)
// or skipped (reduce) — no 64-bit divides anywhere in the element loop
// (the first ring implementation's per-vector i/cvec, %HpWp, /Wp chains
// were 64-bit software divisions and dominated the padded step).

__global__ __launch_bounds__(BLOCK) void bn_act_fwd_ring_kernel(
    const __bf16* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, __bf16* __restrict__ y, int NROWS,
    int Hp, int Wp, int C, int act, int ring, int cvec_sh) {
  const int cvec = 1 << cvec_sh;
  const int rowvecs = Wp * C / 8;
  const int c0 = (threadIdx.x & (cvec - 1)) * 8;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[c0 + j];
    sh[j] = shift[c0 + j];
  }
  for (int r = blockIdx.x; r < NROWS; r += gridDim.x) {
    const int n = r / Hp;                 // one scalar divide per row
    const int yy = r - n * Hp;
    const bool rrow = yy < ring || yy >= Hp - ring;
    const long base = ((long)r * Wp) * cvec;   // row base in vec units
    for (int j = threadIdx.x; j < rowvecs; j += BLOCK) {
      const long i = base + j;
      const int xpix = j >> cvec_sh;
      if (rrow || xpix < ring || xpix >= Wp - ring) {
        *reinterpret_cast<bf16x8*>(y + i * 8) = bf16x8{};
        continue;
      }
      bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
      bf16x8 o;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float vv = (float)v[k] * sc[k] + sh[k];
        o[k] = (__bf16)act_fwd(vv, act);
      }
      *reinterpret_cast<bf16x8*>(y + i * 8) = o;
    }
  }
}

// s1[c] = sum dy', s2[c] = sum dy'*xhat with dy' = dy * dact(y).
// C/8 divides BLOCK and the grid stride, so every thread's channel group is
// FIXED across its whole grid-stride walk: partial sums live in 16 registers
// and each thread issues exactly 16 atomics at the end.
__global__ __launch_bounds__(BLOCK) void bn_act_bwd_reduce_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ rows,  // (gridDim.x, 2, C) per-block partials (stores)
    long nvec, int C, int act) {
  const int cvec = C / 8;
  const int c0 = (int)(((long)blockIdx.x * BLOCK + threadIdx.x) % cvec) * 8;
  float p1[8], p2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) p1[j] = p2[j] = 0.f;
  float mn[8], is[8], ga[8], be[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mn[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
    ga[j] = gamma[c0 + j];
    be[j] = beta[c0 + j];
  }

  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i * 8);
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xhat = ((float)xv[j] - mn[j]) * is[j];
      const float dyp =
          (float)gv[j] * act_bwd_from_u(ga[j] * xhat + be[j], act);
      p1[j] += dyp;
      p2[j] += dyp * xhat;
    }
  }

  // deterministic three-level combine, no atomics anywhere:
  // (1) shfl-reduce lanes sharing a channel group within the wave (fixed
  // order), (2) each wave STORES its totals into its own LDS slab, (3) a
  // serial walk over the slabs writes this block's partial ROW to global;
  // the serial row walk in bn_red_combine finishes the reduction.
  const int lane = threadIdx.x & 63;
  extern __shared__ float sred[];  // (BLOCK/64) * 2*C floats
  const int wave = threadIdx.x >> 6;
  if (cvec < 64) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      for (int off = 32; off >= cvec; off >>= 1) {
        p1[j] += __shfl_down(p1[j], off, 64);
        p2[j] += __shfl_down(p2[j], off, 64);
      }
    }
  }
  if (lane < min(cvec, 64)) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sred[wave * 2 * C + c0 + j] = p1[j];
      sred[wave * 2 * C + C + c0 + j] = p2[j];
    }
  }
  __syncthreads();
  float* row = rows + (long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < 2 * C; i += BLOCK) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < BLOCK / 64; ++w) t += sred[w * 2 * C + i];
    row[i] = t;
  }
}

// ring row-walk reduce: block grid-strides over padded rows, skipping ring
// positions with shift/mask tests only; per-block partial row stores (same
// deterministic fold/combine as the dense path).
__global__ __launch_bounds__(BLOCK) void bn_act_bwd_reduce_ring_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ rows, int NROWS, int Hp, int Wp, int C, int act,
    int ring, int cvec_sh) {
  const int cvec = 1 << cvec_sh;
  const int rowvecs = Wp * C / 8;
  const int c0 = (threadIdx.x & (cvec - 1)) * 8;
  float p1[8], p2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) p1[j] = p2[j] = 0.f;
  float mn[8], is[8], ga[8], be[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mn[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
    ga[j] = gamma[c0 + j];
    be[j] = beta[c0 + j];
  }
  for (int r = blockIdx.x; r < NROWS; r += gridDim.x) {
    const int n = r / Hp;
    const int yy = r - n * Hp;
    if (yy < ring || yy >= Hp - ring) continue;
    const long base = ((long)r * Wp) * cvec;
    for (int j = threadIdx.x; j < rowvecs; j += BLOCK) {
      const int xpix = j >> cvec_sh;
      if (xpix < ring || xpix >= Wp - ring) continue;
      const long i = base + j;
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i * 8);
      bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float xhat = ((float)xv[k] - mn[k]) * is[k];
        const float dyp =
            (float)gv[k] * act_bwd_from_u(ga[k] * xhat + be[k], act);
        p1[k] += dyp;
        p2[k] += dyp * xhat;
      }
    }
  }
  const int lane = threadIdx.x & 63;
  extern __shared__ float sred[];
  const int wave = threadIdx.x >> 6;
  if (cvec < 64) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      for (int off = 32; off >= cvec; off >>= 1) {
        p1[j] += __shfl_down(p1[j], off, 64);
        p2[j] += __shfl_down(p2[j], off, 64);
      }
    }
  }
  if (lane < min(cvec, 64)) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sred[wave * 2 * C + c0 + j] = p1[j];
      sred[wave * 2 * C + C + c0 + j] = p2[j];
    }
  }
  __syncthreads();
  float* row = rows + (long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < 2 * C; i += BLOCK) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < BLOCK / 64; ++w) t += sred[w * 2 * C + i];
    row[i] = t;
  }
}

__global__ __launch_bounds__(BLOCK) void bn_act_bwd_apply_ring_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ scale, const float* __restrict__ red,
    __bf16* __restrict__ dx, int NROWS, int Hp, int Wp, int C, int act,
    float inv_count, int ring, int cvec_sh) {
  const int cvec = 1 << cvec_sh;
  const int rowvecs = Wp * C / 8;
  const int c0 = (threadIdx.x & (cvec - 1)) * 8;
  float mn[8], is[8], ga[8], be[8], sc[8], r1[8], r2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mn[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
    ga[j] = gamma[c0 + j];
    be[j] = beta[c0 + j];
    sc[j] = scale[c0 + j];
    r1[j] = red[c0 + j] * inv_count;
    r2[j] = red[C + c0 + j] * inv_count;
  }
  for (int r = blockIdx.x; r < NROWS; r += gridDim.x) {
    const int n = r / Hp;
    const int yy = r - n * Hp;
    const bool rrow = yy < ring || yy >= Hp - ring;
    const long base = ((long)r * Wp) * cvec;
    for (int j = threadIdx.x; j < rowvecs; j += BLOCK) {
      const long i = base + j;
      const int xpix = j >> cvec_sh;
      if (rrow || xpix < ring || xpix >= Wp - ring) {
        // the previous conv's dgrad GATHERS dx's ring: must be zero
        *reinterpret_cast<bf16x8*>(dx + i * 8) = bf16x8{};
        continue;
      }
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i * 8);
      bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
      bf16x8 o;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float xhat = ((float)xv[k] - mn[k]) * is[k];
        const float dyp =
            (float)gv[k] * act_bwd_from_u(ga[k] * xhat + be[k], act);
        o[k] = (__bf16)(sc[k] * (dyp - r1[k] - xhat * r2[k]));
      }
      *reinterpret_cast<bf16x8*>(dx + i * 8) = o;
    }
  }
}

// red_final[i] = sum over rows (serial -> deterministic); optionally
// accumulates dgamma (= s2) / dbeta (= s1) into caller-owned fp32 buffers
// (the BN params' .grad) so autograd-side accumulation adds disappear.
__global__ void bn_red_combine_kernel(const float* __restrict__ rows,
                                      int nrows, float* __restrict__ red_final,
                                      float* __restrict__ dgamma,
                                      float* __restrict__ dbeta, int C,
                                      int accumulate) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < 2 * C;
       i += gridDim.x * blockDim.x) {
    float t = 0.f;
    for (int r = 0; r < nrows; ++r) t += rows[(long)r * 2 * C + i];
    red_final[i] = t;
    if (i < C) {
      if (dbeta != nullptr) dbeta[i] = accumulate ? dbeta[i] + t : t;
    } else {
      if (dgamma != nullptr)
        dgamma[i - C] = accumulate ? dgamma[i - C] + t : t;
    }
  }
}

// Per-channel sum of an NHWC bf16 tensor (bias gradient: db = sum dy).
// Same fixed-channel-group register walk + shfl/LDS/one-atomic ladder as the
// BN reduce above; ATen's strided (0,2,3) reduction on channels_last runs
// ~8x off bandwidth on these shapes.
__global__ __launch_bounds__(BLOCK) void channel_sum_kernel(
    const __bf16* __restrict__ x, float* __restrict__ rows, long nvec, int C) {
  const int cvec = C / 8;
  const int c0 = (int)(((long)blockIdx.x * BLOCK + threadIdx.x) % cvec) * 8;
  float p[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) p[j] = 0.f;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) p[j] += (float)v[j];
  }
  const int lane = threadIdx.x & 63;
  extern __shared__ float sred[];  // (BLOCK/64) * C floats
  const int wave = threadIdx.x >> 6;
  if (cvec < 64) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      for (int off = 32; off >= cvec; off >>= 1) {
        p[j] += __shfl_down(p[j], off, 64);
      }
    }
  }
  if (lane < min(cvec, 64)) {
#pragma unroll
    for (int j = 0; j < 8; ++j) sred[wave * C + c0 + j] = p[j];
  }
  __syncthreads();
  for (int i = threadIdx.x; i < C; i += BLOCK) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < BLOCK / 64; ++w) t += sred[w * C + i];
    rows[(long)blockIdx.x * C + i] = t;
  }
}

// out[c] (+)= serial sum over per-block partial rows (deterministic).
__global__ void rows_combine_kernel(const float* __restrict__ rows, int nrows,
                                    float* __restrict__ out, long C,
                                    int accumulate) {
  for (long c = (long)blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += (long)gridDim.x * blockDim.x) {
    float t = 0.f;
    for (int r = 0; r < nrows; ++r) t += rows[(long)r * C + c];
    out[c] = accumulate ? out[c] + t : t;
  }
}

// Small-C channel sum (C <= 4: the decoder's nc-channel output, where ATen's
// strided reduce costs ~320us on a 6 MB tensor). Pixel-major scalar walk,
// C fp32 register partials, full-wave shfl reduce, one atomic per block.
template <int C>
__global__ __launch_bounds__(BLOCK) void channel_sum_smallc_kernel(
    const __bf16* __restrict__ x, float* __restrict__ rows, long npix) {
  float p[C];
#pragma unroll
  for (int c = 0; c < C; ++c) p[c] = 0.f;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < npix;
       i += (long)gridDim.x * BLOCK) {
    const __bf16* px = x + i * C;
#pragma unroll
    for (int c = 0; c < C; ++c) p[c] += (float)px[c];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
#pragma unroll
    for (int c = 0; c < C; ++c) p[c] += __shfl_down(p[c], off, 64);
  }
  __shared__ float sw[BLOCK / 64][C];
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
#pragma unroll
    for (int c = 0; c < C; ++c) sw[wave][c] = p[c];
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int c = 0; c < C; ++c) {
      float t = 0.f;
#pragma unroll
      for (int w = 0; w < BLOCK / 64; ++w) t += sw[w][c];
      rows[(long)blockIdx.x * C + c] = t;
    }
  }
}

// dx = scale[c] * (dy' - s1/cnt - xhat * s2/cnt)
__global__ __launch_bounds__(BLOCK) void bn_act_bwd_apply_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ scale, const float* __restrict__ red,
    __bf16* __restrict__ dx, long nvec, int C, int act, float inv_count) {
  const int cvec = C / 8;
  const int c0 = (int)(((long)blockIdx.x * BLOCK + threadIdx.x) % cvec) * 8;
  float mn[8], is[8], ga[8], be[8], sc[8], r1[8], r2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mn[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
    ga[j] = gamma[c0 + j];
    be[j] = beta[c0 + j];
    sc[j] = scale[c0 + j];
    r1[j] = red[c0 + j] * inv_count;
    r2[j] = red[C + c0 + j] * inv_count;
  }
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i * 8);
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xhat = ((float)xv[j] - mn[j]) * is[j];
      const float dyp =
          (float)gv[j] * act_bwd_from_u(ga[j] * xhat + be[j], act);
      const float v = sc[j] * (dyp - r1[j] - xhat * r2[j]);
      o[j] = (__bf16)v;
    }
    *reinterpret_cast<bf16x8*>(dx + i * 8) = o;
  }
}

int pick_grid(long nvec) {
  return (int)std::min<long>(2048, (nvec + BLOCK - 1) / BLOCK);
}

// fold rows [p*64, min(p*64+64, R)) of a (R, C) fp32 matrix into partial row
// p — serial within the chunk, chunks independent: a fixed-structure
// deterministic tree level (the serial flat walk was 11 ms at R=28672).
__global__ void fold_rows_kernel(const float* __restrict__ rows,
                                 float* __restrict__ out, int R, long C) {
  const int p = blockIdx.y;
  const int b0 = p * 64;
  const int b1 = min(b0 + 64, R);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < C;
       i += (long)gridDim.x * blockDim.x) {
    float t = 0.f;
    for (int b = b0; b < b1; ++b) t += rows[(long)b * C + i];
    out[(long)p * C + i] = t;
  }
}

// deterministically reduce a (R, C) partial matrix to (<=64, C): repeated
// 64-way folds, each a fixed association order -> bitwise-stable.
torch::Tensor reduce_rows_det(torch::Tensor rows, long C) {
  auto stream = at::cuda::getCurrentCUDAStream();
  int R = (int)(rows.numel() / C);
  while (R > 64) {
    const int R2 = ceil_div(R, 64);
    auto next = torch::empty({R2, C}, rows.options());
    dim3 grid((unsigned)std::min<long>(256, ceil_div(C, 256)), (unsigned)R2);
    hipLaunchKernelGGL(fold_rows_kernel, grid, dim3(256), 0, stream,
                       rows.data_ptr<float>(), next.data_ptr<float>(), R, C);
    rows = next;
    R = R2;
  }
  return rows;
}

}  // namespace

// Training fwd: x conv output (already stats-accumulated if stats given,
// else stats computed here via torch ops on the wrapper side).
// Returns (y, mean, invstd, scale). shift is internal.
std::vector<torch::Tensor> bn_act_fwd_train(
    torch::Tensor x, torch::Tensor stats, torch::Tensor gamma,
    torch::Tensor beta, c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var, double momentum, double eps,
    long act, long ring) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = x.size(1);
  TORCH_CHECK(C % 8 == 0, "bn_act: C must be a multiple of 8");
  const int Hp = x.size(2), Wp = x.size(3);
  const long count =
      (long)x.size(0) * (Hp - 2 * ring) * (Wp - 2 * ring);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto invstd = torch::empty({C}, f32);
  auto scale = torch::empty({C}, f32);
  auto shift = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();

  int nbuckets = stats.dim() == 3 ? (int)stats.size(0) : 1;
  if (nbuckets > 64) {
    stats = reduce_rows_det(stats, 2L * C);
    nbuckets = (int)stats.size(0);
  }
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(ceil_div(C, 256)), dim3(256), 0,
                     stream, stats.data_ptr<float>(), gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(),
                     running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
                     running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
                     (float)momentum, (float)eps, (float)count, C, nbuckets);

  const long nvec = x.numel() / 8;
  if (ring > 0) {
    const int NROWS = (int)x.size(0) * Hp;
    hipLaunchKernelGGL(bn_act_fwd_ring_kernel,
                       dim3(std::min(2048, NROWS)), dim3(BLOCK), 0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(y.data_ptr()), NROWS, Hp, Wp,
                       C, (int)act, (int)ring, __builtin_ctz(C / 8));
  } else {
    hipLaunchKernelGGL(bn_act_fwd_kernel, dim3(pick_grid(nvec)), dim3(BLOCK),
                       0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(y.data_ptr()), nvec, C,
                       (int)act, Hp, Wp, 0);
  }
  return {y, mean, invstd, scale};
}

torch::Tensor bn_act_fwd_eval(torch::Tensor x, torch::Tensor gamma,
                              torch::Tensor beta, torch::Tensor running_mean,
                              torch::Tensor running_var, double eps, long act,
                              long ring) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = x.size(1);
  TORCH_CHECK(C % 8 == 0, "bn_act: C must be a multiple of 8");
  const int Hp = x.size(2), Wp = x.size(3);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto invstd = torch::empty({C}, f32);
  auto scale = torch::empty({C}, f32);
  auto shift = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(bn_eval_prep_kernel, dim3(ceil_div(C, 256)), dim3(256), 0,
                     stream, running_mean.data_ptr<float>(),
                     running_var.data_ptr<float>(), gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(), (float)eps, C);
  const long nvec = x.numel() / 8;
  if (ring > 0) {
    const int NROWS = (int)x.size(0) * Hp;
    hipLaunchKernelGGL(bn_act_fwd_ring_kernel,
                       dim3(std::min(2048, NROWS)), dim3(BLOCK), 0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(y.data_ptr()), NROWS, Hp, Wp,
                       C, (int)act, (int)ring, __builtin_ctz(C / 8));
  } else {
    hipLaunchKernelGGL(bn_act_fwd_kernel, dim3(pick_grid(nvec)), dim3(BLOCK),
                       0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(y.data_ptr()), nvec, C,
                       (int)act, Hp, Wp, 0);
  }
  return y;
}

// Backward: returns (dx, dgamma, dbeta). Reads only x and dy (activation
// derivative recomputed from the pre-activation value). Deterministic by
// construction (per-block partial rows + serial combine). When
// dgamma_acc/dbeta_acc are given (the BN params' .grad), the combine
// ACCUMULATES into them in place and returns them.
std::vector<torch::Tensor> bn_act_bwd(torch::Tensor x, torch::Tensor dy,
                                      torch::Tensor mean, torch::Tensor invstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      torch::Tensor scale, long act,
                                      c10::optional<torch::Tensor> dgamma_acc,
                                      c10::optional<torch::Tensor> dbeta_acc,
                                      long ring) {
  const int C = x.size(1);
  const int Hp = x.size(2), Wp = x.size(3);
  const long count =
      (long)x.size(0) * (Hp - 2 * ring) * (Wp - 2 * ring);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dx = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  const long nvec = x.numel() / 8;
  const int NROWS = (int)x.size(0) * Hp;
  const int rgrid =
      ring > 0 ? std::min(2048, NROWS) : pick_grid(nvec);
  auto rows = torch::empty({rgrid, 2, C}, f32);     // stores — never zeroed
  auto red = torch::empty({2, C}, f32);
  if (ring > 0) {
    hipLaunchKernelGGL(bn_act_bwd_reduce_ring_kernel, dim3(rgrid), dim3(BLOCK),
                       (BLOCK / 64) * 2 * C * sizeof(float), stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       rows.data_ptr<float>(), NROWS, Hp, Wp, C, (int)act,
                       (int)ring, __builtin_ctz(C / 8));
  } else {
    hipLaunchKernelGGL(bn_act_bwd_reduce_kernel, dim3(rgrid), dim3(BLOCK),
                       (BLOCK / 64) * 2 * C * sizeof(float), stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       rows.data_ptr<float>(), nvec, C, (int)act);
  }
  const int accumulate = dgamma_acc.has_value() ? 1 : 0;
  torch::Tensor dgamma =
      accumulate ? dgamma_acc.value() : torch::empty({C}, f32);
  torch::Tensor dbeta = accumulate ? dbeta_acc.value() : torch::empty({C}, f32);
  torch::Tensor folded = rows;
  int nrows = rgrid;
  if (nrows > 64) {
    folded = reduce_rows_det(rows, 2L * C);
    nrows = (int)folded.size(0);
  }
  hipLaunchKernelGGL(bn_red_combine_kernel, dim3(ceil_div(2 * C, 256)),
                     dim3(256), 0, stream, folded.data_ptr<float>(), nrows,
                     red.data_ptr<float>(), dgamma.data_ptr<float>(),
                     dbeta.data_ptr<float>(), C, accumulate);
  if (ring > 0) {
    hipLaunchKernelGGL(bn_act_bwd_apply_ring_kernel,
                       dim3(std::min(2048, NROWS)), dim3(BLOCK), 0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       scale.data_ptr<float>(), red.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), NROWS, Hp, Wp,
                       C, (int)act, (float)(1.0 / count), (int)ring,
                       __builtin_ctz(C / 8));
  } else {
    hipLaunchKernelGGL(bn_act_bwd_apply_kernel, dim3(pick_grid(nvec)),
                       dim3(BLOCK), 0, stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       scale.data_ptr<float>(), red.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), nvec, C,
                       (int)act, (float)(1.0 / count));
  }
  return {dx, dgamma, dbeta};
}

// db = per-channel sum of a channels_last bf16 tensor, fp32 out.
// Deterministic (row stores + serial combine). With `acc` (the bias .grad),
// accumulates into it in place and returns it.
torch::Tensor channel_sum_nhwc(torch::Tensor x,
                               c10::optional<torch::Tensor> acc) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "channel_sum_nhwc: need channels_last bf16 CUDA tensor");
  const int C = x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  const int accumulate = acc.has_value() ? 1 : 0;
  torch::Tensor out = accumulate ? acc.value() : torch::empty({C}, f32);
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid;
  torch::Tensor rows;
  if (C <= 4) {
    // decoder output channels (nc = 1 or 3): pixel-major scalar walk
    const long npix = x.numel() / C;
    grid = (int)std::min<long>(2048, (npix + BLOCK - 1) / BLOCK);
    rows = torch::empty({grid, C}, f32);
    const __bf16* px = reinterpret_cast<const __bf16*>(x.data_ptr());
    float* po = rows.data_ptr<float>();
    switch (C) {
      case 1: hipLaunchKernelGGL((channel_sum_smallc_kernel<1>), dim3(grid),
                                 dim3(BLOCK), 0, stream, px, po, npix); break;
      case 2: hipLaunchKernelGGL((channel_sum_smallc_kernel<2>), dim3(grid),
                                 dim3(BLOCK), 0, stream, px, po, npix); break;
      case 3: hipLaunchKernelGGL((channel_sum_smallc_kernel<3>), dim3(grid),
                                 dim3(BLOCK), 0, stream, px, po, npix); break;
      default: hipLaunchKernelGGL((channel_sum_smallc_kernel<4>), dim3(grid),
                                  dim3(BLOCK), 0, stream, px, po, npix);
    }
  } else {
    TORCH_CHECK(C % 8 == 0 && 256 % (C / 8) == 0,
                "channel_sum_nhwc: C/8 must divide 256 (or C <= 4)");
    const long nvec = x.numel() / 8;
    grid = pick_grid(nvec);
    rows = torch::empty({grid, C}, f32);
    hipLaunchKernelGGL(channel_sum_kernel, dim3(grid), dim3(BLOCK),
                       (BLOCK / 64) * C * sizeof(float), stream,
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       rows.data_ptr<float>(), nvec, C);
  }
  if (grid > 64) {
    rows = reduce_rows_det(rows, (long)C);
    grid = (int)rows.size(0);
  }
  hipLaunchKernelGGL(rows_combine_kernel,
                     dim3((int)std::max(1L, std::min(32L, (long)ceil_div(C, 256)))),
                     dim3(256), 0, stream, rows.data_ptr<float>(), grid,
                     out.data_ptr<float>(), (long)C, accumulate);
  return out;
}
