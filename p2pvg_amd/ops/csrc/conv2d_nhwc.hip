// NHWC implicit-GEMM convolution on MFMA for gfx950.
//
// Covers the conv workloads of SURVEY §2.6 K1-K3 (dispatch sites
// reference models/vgg_64.py:8 k3s1p1, models/dcgan_64.py:8 k4s2p1,
// decoder ConvTranspose k4s2p1):
//   out[n,yo,xo,k] = act(bias[k] + sum_{r,s,c} in[n, ho*S-P+r, wo*S-P+s, c]
//                                              * w[k,r,s,c])
//
// Three staging modes share one MFMA core:
// - dense (r,s)-outer / 64-wide c-chunk-inner: the im2col gather of a chunk
//   is a CONTIGUOUS 128-byte NHWC run per pixel.
// - RSCLIN (C < 32, e.g. the nc=1/3 first layers): chunks walk the rsc index
//   linearly so a chunk packs many taps — no zero-padding waste; weights stay
//   contiguous, input gathers go per-element.
// - FRAC (fractionally-strided): grid.z = stride^2 parities, each with its
//   own tap map (r,s taps valid for that output parity) and scatter offsets;
//   realizes ConvTranspose fwd / stride-2 dgrad as dense stride-1
//   sub-convolutions in ONE launch — no dilated intermediates, no transposes,
//   no host-side weight slicing.
//
// Tiling: template (BM x BN) block tile over (pixels x channels), 4 waves as
// 2x2, v_mfma_f32_16x16x32_bf16 with fp32 accumulation, LDS tiles
// XOR-swizzled (byte ^= (row&7)<<4) for <=2-way ds_read_b128 conflicts
// (guide §6 G4 / T2). Epilogue fuses bias + activation (K5) and optional
// per-channel sum/sumsq accumulation for downstream BatchNorm (K4).
// blockIdx.x is XCD-swizzled over M (T1).

#include "common.h"

#include <cstdlib>

// glds 3-buffer pipeline for the in-bounds (pad=0 / padded-input) case
std::vector<torch::Tensor> conv2d_glds_fwd(
    torch::Tensor in, torch::Tensor w, c10::optional<torch::Tensor> bias,
    long stride, long act, bool want_stats, long oh, long ow, long oy0,
    long ox0, c10::optional<torch::Tensor> in2,
    c10::optional<torch::Tensor> out2_k1);
bool conv2d_glds_eligible(long C, long K, long R, long stride, long pad);

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

bool glds_enabled() {
  static int v = [] {
    const char* e = std::getenv("P2PVG_GLDS");
    return (e != nullptr && e[0] == '0') ? 0 : 1;
  }();
  return v != 0;
}

constexpr int THREADS = 256;

// BK = K-chunk width in bf16 elements (LDS tile row length)
template <int BK>
__device__ __forceinline__ int swz(int row, int cb) {
  return row * (BK * 2) + (cb ^ ((row & 7) << 4));
}

__device__ __forceinline__ float activate(float v, int act) {
  switch (act) {
    case 1: return v > 0.f ? v : 0.2f * v;          // LeakyReLU(0.2)
    case 2: return tanhf(v);
    case 3: return 1.f / (1.f + __expf(-v));
    default: return v;
  }
}

struct ConvArgs {
  int Nb, H, W, C, K;
  int HO, WO;           // compact output grid (per parity when FRAC)
  int OH, OW;           // full output spatial dims
  int act;
  int mblocks;
  int wk;               // full weight kernel size (for FRAC weight indexing)
  // per-parity tap maps and scatter (index [parity][tap]); parity 0 used
  // for the non-FRAC case with identity maps.
  int rmap[4][4];
  int smap[4][4];
  int padh[4], padw[4];
  int oy0[4], ox0[4], oys;
  float* stats;         // optional (2,K) sum/sumsq accumulation, or null
};

// KSIZE: compact kernel size (taps per axis). FRAC: parity mode. RSCLIN:
// rsc-linear chunking for small C.
template <int BM, int BN, int BK, int KSIZE, int STRIDE, bool FRAC, bool RSCLIN>
__global__ __launch_bounds__(THREADS) void conv2d_nhwc_fwd_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ w,
    const float* __restrict__ bias, __bf16* __restrict__ out, ConvArgs a) {
  __shared__ __align__(16) char lds[(BM + BN) * (BK * 2) + BM * 16];
  char* a_lds = lds;
  char* b_lds = lds + BM * (BK * 2);
  long* pix_out = reinterpret_cast<long*>(lds + (BM + BN) * (BK * 2));
  int* pix_off = reinterpret_cast<int*>(pix_out + BM);
  short* pix_hi = reinterpret_cast<short*>(pix_off + BM);
  short* pix_wi = pix_hi + BM;

  const int par = FRAC ? blockIdx.z : 0;

  // XCD-aware bijective remap of the M dimension
  int bm_lin = blockIdx.x;
  {
    const int nxcd = 8;
    const int q = a.mblocks / nxcd, r = a.mblocks % nxcd;
    const int xcd = bm_lin % nxcd, idx = bm_lin / nxcd;
    bm_lin = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bm_lin * BM;
  const int k0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int M = a.Nb * a.HO * a.WO;

  for (int i = tid; i < BM; i += THREADS) {
    const int pix = m0 + i;
    if (pix < M) {
      const int n = pix / (a.HO * a.WO);
      const int rem = pix - n * (a.HO * a.WO);
      const int ho = rem / a.WO;
      const int wo = rem - ho * a.WO;
      pix_off[i] = n * a.H * a.W * a.C;
      pix_hi[i] = (short)(ho * STRIDE - a.padh[par]);
      pix_wi[i] = (short)(wo * STRIDE - a.padw[par]);
      pix_out[i] =
          (((long)n * a.OH + (ho * a.oys + a.oy0[par])) * a.OW +
           (wo * a.oys + a.ox0[par])) *
          a.K;
    } else {
      pix_off[i] = 0;
      pix_hi[i] = (short)-30000;
      pix_wi[i] = (short)-30000;
      pix_out[i] = -1;
    }
  }
  __syncthreads();

  const int wid = tid >> 6;
  const int lane = tid & 63;
  constexpr int FM = BM / 32;        // M fragments per wave (wave tile BM/2)
  constexpr int FN = BN / 32;        // N fragments per wave
  const int wm = (wid >> 1) * (BM / 2);
  const int wn = (wid & 1) * (BN / 2);

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const bool cvec = (a.C % 8) == 0;
  const int RSC = KSIZE * KSIZE * a.C;
  // chunk loop bounds
  const int n_outer = RSCLIN ? 1 : KSIZE * KSIZE;
  const int n_inner = RSCLIN ? (RSC + BK - 1) / BK : (a.C + BK - 1) / BK;
  const int nchunks = n_outer * n_inner;

  constexpr int ASL = (BM * (BK / 8)) / THREADS;   // A staging slots/thread
  constexpr int BSL = (BN * (BK / 8)) / THREADS;   // B staging slots/thread
  bf16x8 areg[ASL], breg[BSL];

  // Each thread's staging rows are FIXED (slot = it*THREADS + tid), so the
  // gather base addresses only change when the (r,s) tap does: cache them in
  // registers across the n_inner C-chunks of a tap (the staging address math
  // was 39% VALUBusy vs 15% MfmaUtil before this).
  long abase[ASL];   // A source element offset at c==0, or -1 out-of-bounds
  long bbase[BSL];   // B (weight) element offset at c==0, or -1 masked row
  int last_tap = -1;

  auto tap_setup = [&](int oidx) {
    const int ro = oidx / KSIZE, so = oidx % KSIZE;
    const int rw = FRAC ? a.rmap[par][ro] : ro;
    const int sw = FRAC ? a.smap[par][so] : so;
#pragma unroll
    for (int it = 0; it < ASL; ++it) {
      const int row = (it * THREADS + tid) / (BK / 8);
      const int hi = pix_hi[row] + ro;
      const int wi = pix_wi[row] + so;
      abase[it] = (hi >= 0 && hi < a.H && wi >= 0 && wi < a.W)
                      ? (long)pix_off[row] + ((long)hi * a.W + wi) * a.C
                      : -1;
    }
#pragma unroll
    for (int it = 0; it < BSL; ++it) {
      const int row = (it * THREADS + tid) / (BK / 8);
      const int k = k0 + row;
      bbase[it] =
          k < a.K ? (((long)k * a.wk + rw) * a.wk + sw) * a.C : -1;
    }
  };

  // T14 software pipeline (guide §6 G15): issue chunk t+1's global loads,
  // compute chunk t from LDS, then after the barrier write t+1's registers
  // and immediately issue t+2 — HBM latency hides under the MFMA phase with
  // a single LDS buffer (34 KB -> 4 blocks/CU occupancy preserved).
  auto load_chunk = [&](int t) {
    const int oidx = t / n_inner;
    const int ci = t - oidx * n_inner;
    const int c0 = ci * BK;
    if (!RSCLIN && oidx != last_tap) {
      tap_setup(oidx);
      last_tap = oidx;
    }
#pragma unroll
    for (int it = 0; it < ASL; ++it) {
      const int slot = it * THREADS + tid;
      const int cb = (slot % (BK / 8)) * 16;
      bf16x8 v = {};
      if (RSCLIN) {
        const int row = slot / (BK / 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int q = c0 + (cb >> 1) + j;
          if (q < RSC) {
            const int rs = q / a.C;
            const int c = q - rs * a.C;
            const int r = rs / KSIZE, sx = rs % KSIZE;
            const int hi = pix_hi[row] + r;
            const int wi = pix_wi[row] + sx;
            if (hi >= 0 && hi < a.H && wi >= 0 && wi < a.W)
              v[j] = in[(long)pix_off[row] + ((long)hi * a.W + wi) * a.C + c];
          }
        }
      } else {
        const int c = c0 + (cb >> 1);
        if (abase[it] >= 0 && c < a.C) {
          const __bf16* src = in + abase[it] + c;
          if (cvec && c + 8 <= a.C) {
            v = *reinterpret_cast<const bf16x8*>(src);
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (c + j < a.C) v[j] = src[j];
          }
        }
      }
      areg[it] = v;
    }
#pragma unroll
    for (int it = 0; it < BSL; ++it) {
      const int slot = it * THREADS + tid;
      const int cb = (slot % (BK / 8)) * 16;
      bf16x8 v = {};
      if (RSCLIN) {
        const int row = slot / (BK / 8);
        const int k = k0 + row;
        if (k < a.K) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int q = c0 + (cb >> 1) + j;
            if (q < RSC) {
              long off;
              if (FRAC) {
                const int rs = q / a.C;
                const int c = q - rs * a.C;
                const int r = a.rmap[par][rs / KSIZE];
                const int sx = a.smap[par][rs % KSIZE];
                off = (((long)k * a.wk + r) * a.wk + sx) * a.C + c;
              } else {
                off = (long)k * RSC + q;
              }
              v[j] = w[off];
            }
          }
        }
      } else {
        const int c = c0 + (cb >> 1);
        if (bbase[it] >= 0 && c < a.C) {
          const __bf16* src = w + bbase[it] + c;
          if (cvec && c + 8 <= a.C) {
            v = *reinterpret_cast<const bf16x8*>(src);
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (c + j < a.C) v[j] = src[j];
          }
        }
      }
      breg[it] = v;
    }
  };

  auto write_chunk = [&]() {
#pragma unroll
    for (int it = 0; it < ASL; ++it) {
      const int slot = it * THREADS + tid;
      *reinterpret_cast<bf16x8*>(
          a_lds + swz<BK>(slot / (BK / 8), (slot % (BK / 8)) * 16)) = areg[it];
    }
#pragma unroll
    for (int it = 0; it < BSL; ++it) {
      const int slot = it * THREADS + tid;
      *reinterpret_cast<bf16x8*>(
          b_lds + swz<BK>(slot / (BK / 8), (slot % (BK / 8)) * 16)) = breg[it];
    }
  };

  // prologue: chunk 0 into LDS, chunk 1 in flight
  load_chunk(0);
  write_chunk();
  if (nchunks > 1) load_chunk(1);
  __syncthreads();

  for (int t = 0; t < nchunks; ++t) {
    // ---- MFMA over the BK-wide chunk (BK/32 x K=32 steps) ----
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      const int cb = kk * 64 + ((lane >> 4) * 16);
      bf16x8 a_frag[FM], b_frag[FN];
#pragma unroll
      for (int f = 0; f < FM; ++f) {
        const int arow = wm + f * 16 + (lane & 15);
        a_frag[f] = *reinterpret_cast<const bf16x8*>(a_lds + swz<BK>(arow, cb));
      }
#pragma unroll
      for (int f = 0; f < FN; ++f) {
        const int brow = wn + f * 16 + (lane & 15);
        b_frag[f] = *reinterpret_cast<const bf16x8*>(b_lds + swz<BK>(brow, cb));
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
    if (t + 1 < nchunks) {
      __syncthreads();          // everyone done reading chunk t
      write_chunk();            // chunk t+1 registers -> LDS
      if (t + 2 < nchunks) load_chunk(t + 2);  // re-issue immediately
      __syncthreads();          // chunk t+1 visible
    }
  }

  // ---- epilogue ----
#pragma unroll
  for (int j = 0; j < FN; ++j) {
    const int col = k0 + wn + j * 16 + (lane & 15);
    const bool colv = col < a.K;
    const float bv = (colv && bias != nullptr) ? bias[col] : 0.f;
    float csum = 0.f, csq = 0.f;
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const int lrow = wm + i * 16 + (lane >> 4) * 4 + v;
        const long ooff = pix_out[lrow];
        if (colv && ooff >= 0) {
          const float val = activate(acc[i][j][v] + bv, a.act);
          out[ooff + col] = (__bf16)val;
          csum += val;
          csq += val * val;
        }
      }
    }
    // all 64 lanes reach the shuffles (no divergent early-out above)
    if (a.stats != nullptr) {
      // reduce the 4 lanes sharing this column (l, l+16, l+32, l+48), then
      // STORE the per-(block, wave-pair) partial into its own bucket row:
      // no atomics, no pre-zeroing, and the serial bucket walk in
      // bn_finalize makes the statistics bitwise-deterministic.
      csum += __shfl_xor(csum, 16, 64);
      csum += __shfl_xor(csum, 32, 64);
      csq += __shfl_xor(csq, 16, 64);
      csq += __shfl_xor(csq, 32, 64);
      if ((lane >> 4) == 0 && colv) {
        // waves (0,1) and (2,3) cover disjoint col halves x disjoint M
        // halves; 2 rows per block make every (row, col) written exactly once
        const long row =
            ((long)blockIdx.z * a.mblocks + blockIdx.x) * 2 + (wid >> 1);
        float* slot = a.stats + row * 2 * a.K;
        slot[col] = csum;
        slot[a.K + col] = csq;
      }
    }
  }
}

template <int BM, int BN, int BK, int KSIZE, int STRIDE, bool FRAC, bool RSCLIN>
void launch_one(const torch::Tensor& in, const torch::Tensor& w,
                const c10::optional<torch::Tensor>& bias, torch::Tensor& out,
                ConvArgs& a, int nz, bool want_stats,
                torch::Tensor& stats_out) {
  const int M = a.Nb * a.HO * a.WO;
  a.mblocks = ceil_div(M, BM);
  if (want_stats) {
    // one bucket row per (block, wave-pair): written by STORES in the
    // epilogue — never zeroed, combined serially in bn_finalize
    stats_out = torch::empty({(long)a.mblocks * nz * 2, 2, a.K},
                             in.options().dtype(torch::kFloat32));
    a.stats = stats_out.data_ptr<float>();
  }
  dim3 grid(a.mblocks, ceil_div(a.K, BN), nz);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      (conv2d_nhwc_fwd_kernel<BM, BN, BK, KSIZE, STRIDE, FRAC, RSCLIN>), grid,
      dim3(THREADS), 0, stream,
      reinterpret_cast<const __bf16*>(in.data_ptr()),
      reinterpret_cast<const __bf16*>(w.data_ptr()),
      bias.has_value() ? bias->data_ptr<float>() : nullptr,
      reinterpret_cast<__bf16*>(out.data_ptr()), a);
}

// choose tile by problem size, then kernel geometry
template <int KSIZE, int STRIDE, bool FRAC>
void dispatch_tile(const torch::Tensor& in, const torch::Tensor& w,
                   const c10::optional<torch::Tensor>& bias,
                   torch::Tensor& out, ConvArgs& a, int nz, bool want_stats,
                   torch::Tensor& stats_out) {
  const bool rsclin = !FRAC && a.C < 32 && KSIZE > 1;
  const int M = a.Nb * a.HO * a.WO;
  const bool small = (long)ceil_div(M, 128) * ceil_div(a.K, 128) < 160;
  const bool narrow = a.K <= 64;  // half a BN=128 tile would be masked out
  // BK=128 was measured 20-60% SLOWER on the deep-C shapes (64 KB LDS cuts
  // occupancy 4->2 blocks/CU; the barrier savings don't cover it — matches
  // the guide's BK=128 regression note). Keep BK=64 everywhere.
  const bool deep = false;
  if (rsclin) {
    if (narrow)
      launch_one<128, 64, 64, KSIZE, STRIDE, FRAC, true>(in, w, bias, out, a, nz, want_stats, stats_out);
    else
      launch_one<128, 128, 64, KSIZE, STRIDE, FRAC, true>(in, w, bias, out, a, nz, want_stats, stats_out);
  } else if (small) {
    launch_one<64, 64, 64, KSIZE, STRIDE, FRAC, false>(in, w, bias, out, a, nz, want_stats, stats_out);
  } else if (narrow) {
    if (deep)
      launch_one<128, 64, 128, KSIZE, STRIDE, FRAC, false>(in, w, bias, out, a, nz, want_stats, stats_out);
    else
      launch_one<128, 64, 64, KSIZE, STRIDE, FRAC, false>(in, w, bias, out, a, nz, want_stats, stats_out);
  } else if (deep) {
    launch_one<128, 128, 128, KSIZE, STRIDE, FRAC, false>(in, w, bias, out, a, nz, want_stats, stats_out);
  } else {
    launch_one<128, 128, 64, KSIZE, STRIDE, FRAC, false>(in, w, bias, out, a, nz, want_stats, stats_out);
  }
}

void check_nhwc_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16,
              name, " must be a bf16 GPU tensor");
  TORCH_CHECK(t.is_contiguous(at::MemoryFormat::ChannelsLast),
              name, " must be channels_last");
}

}  // namespace

// in: channels_last (N,C,H,W) bf16; w: channels_last (K,C,R,S) bf16;
// bias: (K) fp32 optional. act: 0 none, 1 leaky(0.2), 2 tanh, 3 sigmoid.
// stats: optional (2,K) fp32 ZEROED buffer accumulating sum/sumsq of the
// activated output per channel (for fused BatchNorm statistics).
std::vector<torch::Tensor> conv2d_nhwc_fwd(torch::Tensor in, torch::Tensor w,
                                            c10::optional<torch::Tensor> bias,
                                            long stride, long pad, long act,
                                            bool want_stats, long oh, long ow,
                                            long oy0, long ox0) {
  check_nhwc_bf16(in, "in");
  check_nhwc_bf16(w, "w");
  const int Nb = in.size(0), C = in.size(1), H = in.size(2), W = in.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C, "channel mismatch");
  TORCH_CHECK(R == S, "square kernels only");
  const int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  const int WO = (W + 2 * (int)pad - R) / (int)stride + 1;

  if (conv2d_glds_eligible(C, K, R, stride, pad) &&
      ((long)Nb * HO * WO / 256) * (K / (K % 128 ? 64 : 128)) >= 200 &&
      glds_enabled()) {
    return conv2d_glds_fwd(in, w, bias, stride, act, want_stats, oh, ow, oy0,
                           ox0, c10::nullopt, c10::nullopt);
  }

  const int OH = oh > 0 ? (int)oh : HO;
  const int OW = ow > 0 ? (int)ow : WO;
  auto out = torch::empty({Nb, K, OH, OW},
                          in.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (bias.has_value()) {
    CHECK_INPUT(bias.value());
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be fp32");
  }
  ConvArgs a{};
  a.Nb = Nb; a.H = H; a.W = W; a.C = C; a.K = K;
  a.HO = HO; a.WO = WO; a.OH = OH; a.OW = OW;
  a.act = (int)act; a.wk = R; a.oys = 1;
  a.oy0[0] = (int)oy0; a.ox0[0] = (int)ox0;
  a.padh[0] = (int)pad; a.padw[0] = (int)pad;
  a.stats = nullptr;
  torch::Tensor stats_out;

  if (R == 3 && stride == 1) {
    dispatch_tile<3, 1, false>(in, w, bias, out, a, 1, want_stats, stats_out);
  } else if (R == 4 && stride == 2) {
    dispatch_tile<4, 2, false>(in, w, bias, out, a, 1, want_stats, stats_out);
  } else if (R == 4 && stride == 1) {
    dispatch_tile<4, 1, false>(in, w, bias, out, a, 1, want_stats, stats_out);
  } else if (R == 2 && stride == 1) {
    dispatch_tile<2, 1, false>(in, w, bias, out, a, 1, want_stats, stats_out);
  } else if (R == 1 && stride == 1) {
    dispatch_tile<1, 1, false>(in, w, bias, out, a, 1, want_stats, stats_out);
  } else {
    TORCH_CHECK(false, "unsupported conv geometry: k=", R, " stride=", stride);
  }
  return {out, want_stats ? stats_out : torch::Tensor()};
}

// Fractionally-strided convolution (ConvTranspose fwd / stride-s dgrad):
// out[n, y, x, k] = sum over taps with y = ho*s + py etc. One launch,
// grid.z = s^2 parities. `w` is the FULL (K, C, wk, wk) channels_last weight
// (already arranged so dim0 = output channels); taps are remapped in-kernel.
// up_pad is the fractional-conv padding (the original conv's pad).
std::vector<torch::Tensor> conv2d_nhwc_fracstride(
    torch::Tensor in, torch::Tensor w, c10::optional<torch::Tensor> bias,
    long up_stride, long up_pad, long OH, long OW, long act,
    bool want_stats, long in_ring, long out_ring) {
  check_nhwc_bf16(in, "in");
  check_nhwc_bf16(w, "w");
  const int Nb = in.size(0), C = in.size(1), H = in.size(2), W = in.size(3);
  const int K = w.size(0), WK = w.size(2);
  TORCH_CHECK(w.size(1) == C, "channel mismatch");
  TORCH_CHECK(up_stride == 2, "fracstride: stride 2 only");
  const int st = (int)up_stride, pad = (int)up_pad;

  auto out = torch::empty(
      {Nb, K, OH + 2 * out_ring, OW + 2 * out_ring},
      in.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (bias.has_value()) {
    CHECK_INPUT(bias.value());
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be fp32");
  }

  ConvArgs a{};
  a.Nb = Nb; a.H = H; a.W = W; a.C = C; a.K = K;
  a.OH = (int)(OH + 2 * out_ring); a.OW = (int)(OW + 2 * out_ring);
  a.act = (int)act; a.wk = WK; a.oys = st;
  a.stats = nullptr;
  torch::Tensor stats_out;

  // taps per parity: r with (p + pad - r) % st == 0, offset (p+pad-r)/st
  int ktaps = -1;
  int HOc = -1, WOc = -1;
  for (int p = 0; p < st; ++p) {
    int taps[8][2], nt = 0;
    for (int r = 0; r < WK; ++r) {
      if ((p + pad - r) % st == 0) {
        taps[nt][0] = r;
        taps[nt][1] = (p + pad - r) / st;
        ++nt;
      }
    }
    // sort by offset ascending
    for (int i = 0; i < nt; ++i)
      for (int j = i + 1; j < nt; ++j)
        if (taps[j][1] < taps[i][1]) {
          std::swap(taps[i][0], taps[j][0]);
          std::swap(taps[i][1], taps[j][1]);
        }
    const int ipad = -taps[0][1];
    for (int i = 0; i < nt; ++i)
      TORCH_CHECK(taps[i][1] == i - ipad, "non-consecutive tap offsets");
    if (ktaps < 0) ktaps = nt;
    TORCH_CHECK(nt == ktaps, "tap count varies across parities (pad=", pad,
                " k=", WK, ")");
    // the same axis data serves y-parity rows and x-parity columns
    for (int px = 0; px < st; ++px) {
      const int pi = p * st + px;
      for (int i = 0; i < nt; ++i) a.rmap[pi][i] = taps[i][0];
      a.padh[pi] = ipad - (int)in_ring;   // padded input: shift gathers +ring
      a.oy0[pi] = p + (int)out_ring;      // padded output: interior origin
    }
    for (int py = 0; py < st; ++py) {
      const int pi = py * st + p;
      for (int i = 0; i < nt; ++i) a.smap[pi][i] = taps[i][0];
      a.padw[pi] = ipad - (int)in_ring;
      a.ox0[pi] = p + (int)out_ring;
    }
    // compact grid dims are the same for all parities when OH even
    // (OH here is the LOGICAL output height; rings only move origins)
    const int hoc = ((int)OH - 1 - p) / st + 1;
    if (HOc < 0) HOc = hoc; else TORCH_CHECK(hoc == HOc, "uneven parity grid");
    const int woc = ((int)OW - 1 - p) / st + 1;
    if (WOc < 0) WOc = woc; else TORCH_CHECK(woc == WOc, "uneven parity grid");
  }
  a.HO = HOc; a.WO = WOc;

  TORCH_CHECK(ktaps == 2, "expected 2 taps per axis for k4s2");
  dispatch_tile<2, 1, true>(in, w, bias, out, a, st * st, want_stats, stats_out);
  return {out, want_stats ? stats_out : torch::Tensor()};
}
