// NHWC implicit-GEMM convolution, glds 3-buffer pipeline (gfx950).
//
// Round-2 redesign of the conv forward for the IN-BOUNDS case (docs/ROADMAP
// item 1): when every im2col gather is guaranteed in-bounds — the caller
// passes a PADDED activation whose zero ring realizes the conv padding, so
// the kernel sees pad=0 — staging runs on `global_load_lds` (direct
// HBM->LDS DMA) instead of the register pipeline of conv2d_nhwc.hip:
//  - frees the ~48 staging/base VGPRs that capped the register kernel at
//    2 waves/SIMD,
//  - 3 LDS buffers with COUNTED s_waitcnt vmcnt(N) and raw s_barrier keep
//    two chunks in flight across barriers (guide: +83% over the serial
//    form on the GEMM microbench; __syncthreads would drain the DMA queue).
// Tile: 256 pixels x BN cols, BK=64, 512 threads (8 waves as 2M x 4N).
// LDS images are lane-linear (glds writes base+lane*16); the T2 XOR swizzle
// moves to the per-lane SOURCE address with the same involution the
// ds_read_b128 side applies (guide rule 21).
//
// Covers SURVEY §2.6 K1/K2's k3s1p1 (VGG chains, via padded producers) and
// k4s2p1 (DCGAN) dense shapes with C%64==0 and K%BN==0; everything else
// stays on conv2d_nhwc.hip.

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int GTHREADS = 512;
constexpr int GBM = 256;   // pixel tile

__device__ __forceinline__ int gswz(int row, int cb) {
  return row * 128 + (cb ^ ((row & 7) << 4));
}

__device__ __forceinline__ float gactivate(float v, int act) {
  switch (act) {
    case 1: return v > 0.f ? v : 0.2f * v;
    case 2: return tanhf(v);
    case 3: return 1.f / (1.f + __expf(-v));
    default: return v;
  }
}

template <int N>
__device__ __forceinline__ void waitcnt_vm() {
  // "memory" clobber: the raw barrier/wait pair is NOT an IR-level memory
  // fence, so without it the compiler may hoist the buffer's ds_reads above
  // the wait (observed as a sporadic large-shape numerics failure)
  asm volatile("s_waitcnt vmcnt(%0)" ::"n"(N) : "memory");
}

__device__ __forceinline__ void barrier_mem() {
  asm volatile("" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

__device__ __forceinline__ void glds16(const __bf16* g, char* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)l, 16, 0, 0);
}

struct GArgs {
  int Nb, H, W, C, K;     // H/W: input dims as passed (padded by caller)
  int HO, WO;             // logical output grid
  int OH, OW;             // physical output dims (>= HO/WO when out padded)
  int oy0, ox0;           // interior origin in the physical output
  int act;
  int mblocks;
  float* stats;           // (mblocks*2, 2, K) per-(block, wave-row) stores
  // dual-source input (skip-concat elimination): channels
  // [0, C1) read from `in`, [C1, C) from `in2` (same N/H/W geometry).
  // C1 % 64 == 0 so a BK=64 chunk never straddles the seam.
  const __bf16* in2;
  int C1;
  // dual-destination output (the dgrad of a cat-consuming conv): columns
  // [0, K1) write to `out`, [K1, K) to `out2` (same N/OH/OW geometry).
  __bf16* out2;
  int K1;
};

// KSIZE/STRIDE compile-time; BN: 64 or 128.
template <int BN, int KSIZE, int STRIDE>
__global__ __launch_bounds__(GTHREADS) void conv2d_glds_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ w,
    const float* __restrict__ bias, __bf16* __restrict__ out, GArgs a) {
  extern __shared__ __align__(16) char lds[];
  constexpr int ABYTES = GBM * 128;
  constexpr int BBYTES = BN * 128;
  constexpr int SLAB = ABYTES + BBYTES;
  long* pix_out = reinterpret_cast<long*>(lds + 3 * SLAB);
  long* pix_base = pix_out + GBM;

  // XCD-aware bijective remap of the M dimension (T1)
  int bm_lin = blockIdx.x;
  {
    const int nxcd = 8;
    const int q = a.mblocks / nxcd, r = a.mblocks % nxcd;
    const int xcd = bm_lin % nxcd, idx = bm_lin / nxcd;
    bm_lin = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bm_lin * GBM;
  const int k0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int M = a.Nb * a.HO * a.WO;

  for (int i = tid; i < GBM; i += GTHREADS) {
    const int pix = m0 + i;
    if (pix < M) {
      const int n = pix / (a.HO * a.WO);
      const int rem = pix - n * (a.HO * a.WO);
      const int ho = rem / a.WO;
      const int wo = rem - ho * a.WO;
      // tap-(0,0) gather origin in PIXELS (channel stride applied at stage
      // time: the two dual-source tensors have different C strides)
      pix_base[i] = ((long)n * a.H + ho * STRIDE) * a.W +
                    (long)(wo * STRIDE);
      pix_out[i] = ((long)n * a.OH + (ho + a.oy0)) * a.OW + (wo + a.ox0);
    } else {
      pix_base[i] = 0;      // safe in-bounds dummy reads
      pix_out[i] = -1;
    }
  }
  __syncthreads();

  const int wv = tid >> 6;
  const int lane = tid & 63;
  const int piece = lane & 7;       // 16B piece within a 128B row
  const int rsub = lane >> 3;       // row within an 8-row glds call

  // per-lane staging source bases (element offsets at tap (0,0), c=0) with
  // the read-side XOR swizzle pre-applied to the SOURCE piece (rule 21)
  constexpr int ACALLS = GBM / 8 / 8;          // 4: rows per wave / 8
  constexpr int BCALLS = BN / 8 / 8;           // 1 or 2
  constexpr int BROWS = BN / 8;                // B rows per wave
  int asrc[ACALLS];
  int arowl[ACALLS];
#pragma unroll
  for (int i = 0; i < ACALLS; ++i) {
    arowl[i] = wv * (GBM / 8) + i * 8 + rsub;
    asrc[i] = ((piece ^ (arowl[i] & 7)) << 3);
  }
  const int RSC = KSIZE * KSIZE * a.C;
  long bsrc[BCALLS];
#pragma unroll
  for (int i = 0; i < BCALLS; ++i) {
    const int brow = wv * BROWS + i * 8 + rsub;
    bsrc[i] = (long)(k0 + brow) * RSC + ((piece ^ (brow & 7)) << 3);
  }
  int abase[ACALLS];
#pragma unroll
  for (int i = 0; i < ACALLS; ++i) abase[i] = (int)pix_base[arowl[i]];

  const int n_inner = a.C >> 6;                 // C / 64 chunks per tap
  const int nchunks = KSIZE * KSIZE * n_inner;

  auto stage = [&](int t, int buf) {
    const int tap = t / n_inner;
    const int c0 = (t - tap * n_inner) << 6;
    const int ro = tap / KSIZE, so = tap % KSIZE;
    // dual-source: the whole 64-wide chunk comes from one tensor
    const bool second = a.in2 != nullptr && c0 >= a.C1;
    const __bf16* src = second ? a.in2 : in;
    const int cs = a.in2 == nullptr ? a.C
                                    : (second ? a.C - a.C1 : a.C1);
    const int cl = second ? c0 - a.C1 : c0;
    const int tappix = ro * a.W + so;
    const long btap = (long)tap * a.C + c0;
    char* abuf = lds + buf * SLAB;
    char* bbuf = abuf + ABYTES;
#pragma unroll
    for (int i = 0; i < ACALLS; ++i) {
      glds16(src + (long)(abase[i] + tappix) * cs + cl + asrc[i],
             abuf + (wv * (GBM / 8) + i * 8) * 128);
    }
#pragma unroll
    for (int i = 0; i < BCALLS; ++i) {
      glds16(w + bsrc[i] + btap, bbuf + (wv * BROWS + i * 8) * 128);
    }
  };

  constexpr int FM = 8;                 // 128 rows / 16 per wave (2M split)
  constexpr int FN = BN / 64;           // 32 or 64 cols / 16 per wave (4N)
  const int wm = (wv >> 2) * 128;
  const int wn = (wv & 3) * (BN / 4);
  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int G = ACALLS + BCALLS;    // glds per wave per chunk

  stage(0, 0);
  if (nchunks > 1) stage(1, 1);
  if (nchunks > 2) stage(2, 2);

  for (int t = 0; t < nchunks; ++t) {
    const int buf = t % 3;
    const char* abuf = lds + buf * SLAB;
    const char* bbuf = abuf + ABYTES;
    // chunk t landed once its own glds completed; with fewer than 2 chunks
    // staged ahead (short K-loops / loop tail), a flat vmcnt(2G) would pass
    // while chunk t is still in flight
    const int ahead = min(nchunks, t + 3) - 1 - t;
    if (ahead >= 2) waitcnt_vm<2 * G>();
    else if (ahead == 1) waitcnt_vm<G>();
    else waitcnt_vm<0>();
    barrier_mem();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int cb = kk * 64 + ((lane >> 4) * 16);
      bf16x8 a_frag[FM], b_frag[FN];
#pragma unroll
      for (int f = 0; f < FM; ++f)
        a_frag[f] = *reinterpret_cast<const bf16x8*>(
            abuf + gswz(wm + f * 16 + (lane & 15), cb));
#pragma unroll
      for (int f = 0; f < FN; ++f)
        b_frag[f] = *reinterpret_cast<const bf16x8*>(
            bbuf + gswz(wn + f * 16 + (lane & 15), cb));
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    barrier_mem();                      // everyone done reading buf
    if (t + 3 < nchunks) stage(t + 3, buf);
  }

  // ---- epilogue: bias + activation + optional per-channel stats stores ----
#pragma unroll
  for (int j = 0; j < FN; ++j) {
    const int col = k0 + wn + j * 16 + (lane & 15);
    const float bv = bias != nullptr ? bias[col] : 0.f;
    // dual-destination: column decides the target tensor (dgrad of a
    // cat-consuming conv splits back into the two operand grads)
    const bool osecond = a.out2 != nullptr && col >= a.K1;
    __bf16* dst = osecond ? a.out2 : out;
    const int ks = a.out2 == nullptr ? a.K : (osecond ? a.K - a.K1 : a.K1);
    const int ocol = osecond ? col - a.K1 : col;
    float csum = 0.f, csq = 0.f;
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const int lrow = wm + i * 16 + (lane >> 4) * 4 + v;
        const long opix = pix_out[lrow];
        if (opix >= 0) {
          const float val = gactivate(acc[i][j][v] + bv, a.act);
          dst[opix * ks + ocol] = (__bf16)val;
          csum += val;
          csq += val * val;
        }
      }
    }
    if (a.stats != nullptr) {
      csum += __shfl_xor(csum, 16, 64);
      csum += __shfl_xor(csum, 32, 64);
      csq += __shfl_xor(csq, 16, 64);
      csq += __shfl_xor(csq, 32, 64);
      if ((lane >> 4) == 0) {
        const long row = (long)blockIdx.x * 2 + (wv >> 2);
        float* slot = a.stats + row * 2 * a.K;
        slot[col] = csum;
        slot[a.K + col] = csq;
      }
    }
  }
}

template <int BN, int KSIZE, int STRIDE>
void glaunch(const torch::Tensor& in, const torch::Tensor& w,
             const c10::optional<torch::Tensor>& bias, torch::Tensor& out,
             GArgs& a, bool want_stats, torch::Tensor& stats_out) {
  const int M = a.Nb * a.HO * a.WO;
  a.mblocks = ceil_div(M, GBM);
  if (want_stats) {
    stats_out = torch::empty({(long)a.mblocks * 2, 2, a.K},
                             in.options().dtype(torch::kFloat32));
    a.stats = stats_out.data_ptr<float>();
  }
  const int shmem = 3 * (GBM * 128 + BN * 128) + GBM * 16;
  auto* kfn = (const void*)&conv2d_glds_kernel<BN, KSIZE, STRIDE>;
  static int configured_bn = 0;
  if (configured_bn != BN + KSIZE * 1000 + STRIDE * 100000) {
    (void)hipFuncSetAttribute(kfn, hipFuncAttributeMaxDynamicSharedMemorySize,
                              shmem);
    configured_bn = BN + KSIZE * 1000 + STRIDE * 100000;
  }
  dim3 grid(a.mblocks, a.K / BN);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((conv2d_glds_kernel<BN, KSIZE, STRIDE>), grid,
                     dim3(GTHREADS), shmem, stream,
                     reinterpret_cast<const __bf16*>(in.data_ptr()),
                     reinterpret_cast<const __bf16*>(w.data_ptr()),
                     bias.has_value() ? bias->data_ptr<float>() : nullptr,
                     reinterpret_cast<__bf16*>(out.data_ptr()), a);
}

}  // namespace

bool conv2d_glds_eligible(long C, long K, long R, long stride, long pad) {
  if (pad != 0) return false;                    // padded-input contract
  if (C % 64 != 0) return false;
  if (K % 64 != 0) return false;
  if (!((R == 3 && stride == 1) || (R == 4 && stride == 2) ||
        (R == 4 && stride == 1) || (R == 2 && stride == 1)))
    return false;
  return true;
}

// in: channels_last (N,C,H,W) bf16 where the conv's zero padding is ALREADY
// materialized as a zero ring (so every gather is in-bounds and pad=0);
// w: channels_last (K,C,R,R) bf16. Returns (out, stats). `oh/ow/oy0/ox0`
// optionally place the logical HOxWO output interior inside a larger
// physical (OH,OW) output (for padded-producer chains); <=0 means dense.
std::vector<torch::Tensor> conv2d_glds_fwd(torch::Tensor in, torch::Tensor w,
                                           c10::optional<torch::Tensor> bias,
                                           long stride, long act,
                                           bool want_stats, long oh, long ow,
                                           long oy0, long ox0,
                                           c10::optional<torch::Tensor> in2,
                                           c10::optional<torch::Tensor> out2_k1) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == torch::kBFloat16 &&
                  in.is_contiguous(at::MemoryFormat::ChannelsLast),
              "glds conv: in must be bf16 channels_last GPU");
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                  w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "glds conv: w must be bf16 channels_last GPU");
  const int Nb = in.size(0);
  const int C1 = in.size(1);
  const int C = in2.has_value() ? C1 + (int)in2->size(1) : C1;
  const int H = in.size(2), W = in.size(3);
  const int K = w.size(0), R = w.size(2);
  if (in2.has_value()) {
    TORCH_CHECK(in2->is_cuda() && in2->scalar_type() == torch::kBFloat16 &&
                    in2->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    in2->size(0) == Nb && in2->size(2) == H &&
                    in2->size(3) == W && C1 % 64 == 0,
                "glds conv: in2 geometry mismatch (C1 must be 64-aligned)");
  }
  TORCH_CHECK(w.size(1) == C && w.size(3) == R, "glds conv: weight mismatch");
  TORCH_CHECK(conv2d_glds_eligible(C, K, R, stride, 0),
              "glds conv: unsupported geometry");
  const int HO = (H - R) / (int)stride + 1;
  const int WO = (W - R) / (int)stride + 1;
  const int OH = oh > 0 ? (int)oh : HO;
  const int OW = ow > 0 ? (int)ow : WO;

  const int Kout = out2_k1.has_value() ? K - (int)out2_k1->size(1) : K;
  auto out = torch::empty(
      {Nb, Kout, OH, OW},
      in.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (bias.has_value()) {
    CHECK_INPUT(bias.value());
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be fp32");
  }
  GArgs a{};
  a.Nb = Nb; a.H = H; a.W = W; a.C = C; a.K = K;
  a.HO = HO; a.WO = WO; a.OH = OH; a.OW = OW;
  a.oy0 = (int)oy0; a.ox0 = (int)ox0;
  a.act = (int)act;
  a.in2 = in2.has_value()
              ? reinterpret_cast<const __bf16*>(in2->data_ptr()) : nullptr;
  a.C1 = C1;
  a.out2 = nullptr;
  a.K1 = K;
  if (out2_k1.has_value()) {
    auto& o2 = out2_k1.value();
    TORCH_CHECK(o2.is_cuda() && o2.scalar_type() == torch::kBFloat16 &&
                    o2.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    o2.size(0) == Nb && (int)o2.size(2) == OH &&
                    (int)o2.size(3) == OW,
                "glds conv: out2 geometry mismatch");
    a.out2 = reinterpret_cast<__bf16*>(o2.data_ptr());
    a.K1 = K - (int)o2.size(1);
    TORCH_CHECK(a.K1 > 0 && a.K1 < K, "glds conv: bad out2 split");
  }
  torch::Tensor stats_out;

  const bool bn64 = (K % 128 != 0);
#define GDISPATCH(KS, ST)                                                  \
  do {                                                                     \
    if (bn64)                                                              \
      glaunch<64, KS, ST>(in, w, bias, out, a, want_stats, stats_out);     \
    else                                                                   \
      glaunch<128, KS, ST>(in, w, bias, out, a, want_stats, stats_out);    \
  } while (0)
  if (R == 3 && stride == 1) GDISPATCH(3, 1);
  else if (R == 4 && stride == 2) GDISPATCH(4, 2);
  else if (R == 4 && stride == 1) GDISPATCH(4, 1);
  else if (R == 2 && stride == 1) GDISPATCH(2, 1);
  else TORCH_CHECK(false, "glds conv: unsupported geometry");
#undef GDISPATCH
  return {out, want_stats ? stats_out : torch::Tensor()};
}
