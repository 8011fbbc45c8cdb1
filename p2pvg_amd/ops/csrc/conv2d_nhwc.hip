// NHWC implicit-GEMM convolution on MFMA for gfx950.
//
// Covers the conv workloads of SURVEY §2.6 K1-K3 (dispatch sites
// reference models/vgg_64.py:8 k3s1p1, models/dcgan_64.py:8 k4s2p1,
// encoder tails k4s1p0, decoder ConvTranspose k4s2p1):
//   out[n,yo,xo,k] = act(bias[k] + sum_{r,s,c} in[n, ho*S-P+r, wo*S-P+s, c]
//                                              * w[k,r,s,c])
// where (yo,xo) = (ho*OYS+OY0, wo*OXS+OX0): OYS=1 is plain convolution;
// OYS=2 with a per-parity weight slice realizes fractionally-strided
// convolution (ConvTranspose fwd / stride-2 dgrad) as 4 dense sub-problems —
// the xGMI-free, transpose-free CDNA4 formulation of K2/K3.
//
// GEMM view: M = N*HO*WO output pixels, Ndim = K output channels,
// Kdim = R*S*C, iterated as (r,s) outer x 64-wide c-chunks inner so the
// im2col gather of one chunk is a CONTIGUOUS 128-byte run of the NHWC input
// per pixel (c fastest) — staged straight into LDS.
//
// Tiling: 128x128 block tile (BM pixels x BN channels), 4 waves as 2x2 of
// 64x64 wave tiles, v_mfma_f32_16x16x32_bf16 with fp32 accumulation, LDS
// tiles XOR-swizzled (byte ^= (row&7)<<4) so the ds_read_b128 fragment reads
// are <=2-way bank conflicted (guide §6 G4 / T2). Epilogue fuses bias +
// activation (none/LeakyReLU(0.2)/Tanh/Sigmoid) — the K5 fusion.
// blockIdx.x is XCD-swizzled over the M dimension (T1).

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 64;              // c-chunk (bf16 elements)
constexpr int THREADS = 256;
constexpr int ROW_BYTES = BK * 2;   // 128 B per LDS tile row

__device__ __forceinline__ int swz(int row, int cb) {
  // byte offset into a [rows][BK] bf16 tile with the (row&7)<<4 XOR swizzle
  return row * ROW_BYTES + (cb ^ ((row & 7) << 4));
}

__device__ __forceinline__ float activate(float v, int act) {
  switch (act) {
    case 1: return v > 0.f ? v : 0.2f * v;          // LeakyReLU(0.2)
    case 2: return tanhf(v);
    case 3: return 1.f / (1.f + __expf(-v));
    default: return v;
  }
}

struct ScatterSpec {
  int OH, OW;    // full output spatial dims
  int oys, oy0, ox0;  // y = ho*oys + oy0, x = wo*oys + ox0
};

// in:  (N, H, W, C) bf16   w: (K, R, S, C) bf16   bias: (K) f32 or null
// out: (N, OH, OW, K) bf16, written at the scattered (y,x) positions
template <int KSIZE, int STRIDE>
__global__ __launch_bounds__(THREADS) void conv2d_nhwc_fwd_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ w,
    const float* __restrict__ bias, __bf16* __restrict__ out,
    int Nb, int H, int W, int C, int K, int HO, int WO, int PADH, int PADW,
    int act, int mblocks, ScatterSpec sc) {
  // LDS: A tile (BM x BK) + B tile (BN x BK), single-buffered, + pixel meta
  __shared__ __align__(16) char lds[(BM + BN) * ROW_BYTES + BM * 16];
  char* a_lds = lds;
  char* b_lds = lds + BM * ROW_BYTES;
  long* pix_out = reinterpret_cast<long*>(lds + (BM + BN) * ROW_BYTES);
  int* pix_off = reinterpret_cast<int*>(pix_out + BM);
  short* pix_hi = reinterpret_cast<short*>(pix_off + BM);
  short* pix_wi = pix_hi + BM;

  // XCD-aware remap of the M dimension (bijective variant, guide §5)
  int bm_lin = blockIdx.x;
  {
    const int nxcd = 8;
    const int q = mblocks / nxcd, r = mblocks % nxcd;
    const int xcd = bm_lin % nxcd, idx = bm_lin / nxcd;
    bm_lin = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bm_lin * BM;
  const int k0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int M = Nb * HO * WO;

  // pixel meta
  for (int i = tid; i < BM; i += THREADS) {
    const int pix = m0 + i;
    if (pix < M) {
      const int n = pix / (HO * WO);
      const int rem = pix - n * (HO * WO);
      const int ho = rem / WO;
      const int wo = rem - ho * WO;
      pix_off[i] = n * H * W * C;
      pix_hi[i] = (short)(ho * STRIDE - PADH);
      pix_wi[i] = (short)(wo * STRIDE - PADW);
      pix_out[i] =
          (((long)n * sc.OH + (ho * sc.oys + sc.oy0)) * sc.OW +
           (wo * sc.oys + sc.ox0)) *
          K;
    } else {
      pix_off[i] = 0;
      pix_hi[i] = (short)-30000;  // always out of bounds -> zero rows
      pix_wi[i] = (short)-30000;
      pix_out[i] = -1;
    }
  }
  __syncthreads();

  const int wid = tid >> 6;         // wave 0..3
  const int lane = tid & 63;
  const int wm = (wid >> 1) * 64;   // wave row base in tile
  const int wn = (wid & 1) * 64;    // wave col base in tile

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const bool cvec = (C % 8) == 0;

  for (int r = 0; r < KSIZE; ++r) {
    for (int s = 0; s < KSIZE; ++s) {
      for (int c0 = 0; c0 < C; c0 += BK) {
        // ---- stage A (input patches) ----
        // 128 rows x 8 slots of 16B; 1024 slots over 256 threads
#pragma unroll
        for (int it = 0; it < (BM * 8) / THREADS; ++it) {
          const int slot = it * THREADS + tid;
          const int row = slot >> 3;
          const int cb = (slot & 7) * 16;         // byte col
          const int c = c0 + (cb >> 1);           // element col base
          const int hi = pix_hi[row] + r;
          const int wi = pix_wi[row] + s;
          bf16x8 v = {};
          if (hi >= 0 && hi < H && wi >= 0 && wi < W && c < C) {
            const __bf16* src = in + (long)pix_off[row] + ((long)hi * W + wi) * C + c;
            if (cvec && c + 8 <= C) {
              v = *reinterpret_cast<const bf16x8*>(src);
            } else {
#pragma unroll
              for (int j = 0; j < 8; ++j)
                if (c + j < C) v[j] = src[j];
            }
          }
          *reinterpret_cast<bf16x8*>(a_lds + swz(row, cb)) = v;
        }
        // ---- stage B (weights) ----
#pragma unroll
        for (int it = 0; it < (BN * 8) / THREADS; ++it) {
          const int slot = it * THREADS + tid;
          const int row = slot >> 3;              // out-channel within tile
          const int cb = (slot & 7) * 16;
          const int c = c0 + (cb >> 1);
          const int k = k0 + row;
          bf16x8 v = {};
          if (k < K && c < C) {
            const __bf16* src = w + (((long)k * KSIZE + r) * KSIZE + s) * C + c;
            if (cvec && c + 8 <= C) {
              v = *reinterpret_cast<const bf16x8*>(src);
            } else {
#pragma unroll
              for (int j = 0; j < 8; ++j)
                if (c + j < C) v[j] = src[j];
            }
          }
          *reinterpret_cast<bf16x8*>(b_lds + swz(row, cb)) = v;
        }
        __syncthreads();

        // ---- MFMA over the 64-wide chunk (2 x K=32 steps) ----
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int cb = kk * 64 + ((lane >> 4) * 16);
          bf16x8 a_frag[4], b_frag[4];
#pragma unroll
          for (int f = 0; f < 4; ++f) {
            const int arow = wm + f * 16 + (lane & 15);
            a_frag[f] = *reinterpret_cast<const bf16x8*>(a_lds + swz(arow, cb));
            const int brow = wn + f * 16 + (lane & 15);
            b_frag[f] = *reinterpret_cast<const bf16x8*>(b_lds + swz(brow, cb));
          }
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
      }
    }
  }

  // ---- epilogue: bias + activation, bf16 store ----
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = k0 + wn + j * 16 + (lane & 15);
    if (col >= K) continue;
    const float bv = bias != nullptr ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const int lrow = wm + i * 16 + (lane >> 4) * 4 + v;
        const long ooff = pix_out[lrow];
        if (ooff >= 0) {
          const float val = activate(acc[i][j][v] + bv, act);
          out[ooff + col] = (__bf16)val;
        }
      }
    }
  }
}

template <int KSIZE, int STRIDE>
void launch_fwd(const torch::Tensor& in, const torch::Tensor& w,
                const c10::optional<torch::Tensor>& bias, torch::Tensor& out,
                int Nb, int H, int W, int C, int K, int HO, int WO, int PADH,
                int PADW, int act, ScatterSpec sc) {
  const int M = Nb * HO * WO;
  const int mblocks = ceil_div(M, BM);
  dim3 grid(mblocks, ceil_div(K, BN));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((conv2d_nhwc_fwd_kernel<KSIZE, STRIDE>), grid,
                     dim3(THREADS), 0, stream,
                     reinterpret_cast<const __bf16*>(in.data_ptr()),
                     reinterpret_cast<const __bf16*>(w.data_ptr()),
                     bias.has_value() ? bias->data_ptr<float>() : nullptr,
                     reinterpret_cast<__bf16*>(out.data_ptr()), Nb, H, W, C, K,
                     HO, WO, PADH, PADW, act, mblocks, sc);
}

void dispatch_fwd(const torch::Tensor& in, const torch::Tensor& w,
                  const c10::optional<torch::Tensor>& bias, torch::Tensor& out,
                  int Nb, int H, int W, int C, int K, int HO, int WO, int R,
                  int stride, int PADH, int PADW, int act, ScatterSpec sc) {
  if (R == 3 && stride == 1) {
    launch_fwd<3, 1>(in, w, bias, out, Nb, H, W, C, K, HO, WO, PADH, PADW, act, sc);
  } else if (R == 4 && stride == 2) {
    launch_fwd<4, 2>(in, w, bias, out, Nb, H, W, C, K, HO, WO, PADH, PADW, act, sc);
  } else if (R == 4 && stride == 1) {
    launch_fwd<4, 1>(in, w, bias, out, Nb, H, W, C, K, HO, WO, PADH, PADW, act, sc);
  } else if (R == 2 && stride == 1) {
    launch_fwd<2, 1>(in, w, bias, out, Nb, H, W, C, K, HO, WO, PADH, PADW, act, sc);
  } else if (R == 1 && stride == 1) {
    launch_fwd<1, 1>(in, w, bias, out, Nb, H, W, C, K, HO, WO, PADH, PADW, act, sc);
  } else {
    TORCH_CHECK(false, "unsupported conv geometry: k=", R, " stride=", stride);
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
torch::Tensor conv2d_nhwc_fwd(torch::Tensor in, torch::Tensor w,
                              c10::optional<torch::Tensor> bias, long stride,
                              long pad, long act) {
  check_nhwc_bf16(in, "in");
  check_nhwc_bf16(w, "w");
  const int Nb = in.size(0), C = in.size(1), H = in.size(2), W = in.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C, "channel mismatch");
  TORCH_CHECK(R == S, "square kernels only");
  const int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  const int WO = (W + 2 * (int)pad - R) / (int)stride + 1;

  auto out = torch::empty({Nb, K, HO, WO},
                          in.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (bias.has_value()) {
    CHECK_INPUT(bias.value());
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be fp32");
  }
  ScatterSpec sc{HO, WO, 1, 0, 0};
  dispatch_fwd(in, w, bias, out, Nb, H, W, C, K, HO, WO, R, (int)stride,
               (int)pad, (int)pad, (int)act, sc);
  return out;
}

// Scatter variant: computes a stride-1 conv of `in` with `w` on a compact
// HOxWO grid and writes results at out[:, :, oy0::oys, ox0::oxs]. `out` is
// preallocated by the caller (one call per parity). ipad is the implicit
// input padding of the compact problem.
void conv2d_nhwc_fwd_scatter(torch::Tensor in, torch::Tensor w,
                             c10::optional<torch::Tensor> bias,
                             torch::Tensor out, long ipad_h, long ipad_w,
                             long oys, long oy0, long ox0, long act) {
  check_nhwc_bf16(in, "in");
  check_nhwc_bf16(w, "w");
  check_nhwc_bf16(out, "out");
  const int Nb = in.size(0), C = in.size(1), H = in.size(2), W = in.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C, "channel mismatch");
  TORCH_CHECK(R == S, "square kernels only");
  const int OH = out.size(2), OW = out.size(3);
  // compact grid dims from the scatter spec
  const int HO = (OH - 1 - (int)oy0) / (int)oys + 1;
  const int WO = (OW - 1 - (int)ox0) / (int)oys + 1;
  TORCH_CHECK(out.size(0) == Nb && out.size(1) == K, "bad out shape");

  ScatterSpec sc{OH, OW, (int)oys, (int)oy0, (int)ox0};
  dispatch_fwd(in, w, bias, out, Nb, H, W, C, K, HO, WO, R, 1, (int)ipad_h,
               (int)ipad_w, (int)act, sc);
}
