// NHWC weight-gradient (wgrad) kernel on MFMA for gfx950.
//
// Generic form covering conv wgrad AND ConvTranspose wgrad (roles swapped):
//   dW[b, r, s, a] = sum_{n,ho,wo} Y[n,ho,wo,b] * X[n, ho*ST-P+r, wo*ST-P+s, a]
// For conv: Y = grad_out, X = input, dW is the (K,R,S,C)-physical weight grad.
// For convT: Y = input, X = grad_out (the larger map), giving the
// (Ci,R,S,Co)-physical transposed-conv weight grad.
//
// GEMM view per (r,s): (B x A) = Y^T @ X_patch, reduced over all N*HO*WO
// pixels — a huge-K GEMM, split over pixel slabs (grid.z) with fp32
// atomicAdd into a workspace (deterministic mode: splitP=1).
//
// Both operands are pixel-major in memory (NHWC), so global loads are
// contiguous 16B channel runs and staging writes are single ds_write_b128s
// into [pix/4][ch/16][4][16] subtiles; the transpose to reduction-axis
// fragments happens IN HARDWARE via ds_read_b64_tr_b16 (gfx950's LDS
// transpose read: a 16-lane group reads one 128-byte [4pix][16ch] block and
// lane c receives column c — semantics pinned by tests/test_tr16_gpu.py).

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int PCH = 64;    // pixels per staging chunk
constexpr int THREADS = 256;

typedef unsigned short u16x4 __attribute__((ext_vector_type(4)));

// [pix/4][BT/16][4][16] subtiled LDS image: byte offset of element (pix, ch)
template <int BT>
__device__ __forceinline__ int sub_off(int pix, int ch) {
  return ((pix >> 2) * (BT / 16) + (ch >> 4)) * 128 + (pix & 3) * 32 +
         (ch & 15) * 2;
}

// issue one pair of hardware transpose reads (no wait) for the 8-pixel-run
// fragment of a fixed channel: lane l covers ch (..&15 == l&15), pix pe..pe+7
__device__ __forceinline__ void tr16_issue(const char* tile, int ch, int pe,
                                           int lane, int bt_over_16,
                                           u16x4& lo, u16x4& hi) {
  const unsigned a0 =
      (unsigned)(unsigned long long)tile +
      (unsigned)(((pe >> 2) * bt_over_16 + (ch >> 4)) * 128 + (lane & 15) * 8);
  const unsigned a1 = a0 + (unsigned)(bt_over_16 * 128);
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(lo) : "v"(a0));
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(hi) : "v"(a1));
}

__device__ __forceinline__ bf16x8 tr16_combine(u16x4 lo, u16x4 hi) {
  typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));
  u16x8 r = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
  return __builtin_bit_cast(bf16x8, r);
}

// BTB/BTA: channel tiles per side (64 or 128, independent); wave tile
// (BTB/2 x BTA/2), (BTB/32 x BTA/32) fragments per wave.
// yring > 0: Y is physically (N, HO+2r, WO+2r, B) with the logical content
// in the interior — pixel indices stay over the logical HOxWO grid so no
// FLOPs are spent on the ring.
template <int BTB, int BTA>
__global__ __launch_bounds__(THREADS) void conv2d_wgrad_kernel(
    const __bf16* __restrict__ Y,  // (N, HO, WO, B) (+ ring, see above)
    const __bf16* __restrict__ X,  // (N, H, W, A)
    float* __restrict__ ws,        // (sp, B, R, S, A) fp32 slab stores
    int Nb, int HO, int WO, int B, int H, int W, int A, int R, int S,
    int STRIDE, int PAD, int p_per_slab, int yring) {
  __shared__ __align__(16) char lds[PCH * (BTB + BTA) * 2];
  char* yt = lds;                        // [PCH/4][BTB/16][4][16] subtiles
  char* xt = lds + PCH * BTB * 2;        // [PCH/4][BTA/16][4][16] subtiles

  const int at_blocks = (A + BTA - 1) / BTA;
  const int b0 = (blockIdx.x / at_blocks) * BTB;
  const int a0 = (blockIdx.x % at_blocks) * BTA;
  const int r = blockIdx.y / S;
  const int s = blockIdx.y % S;
  const int p_begin = blockIdx.z * p_per_slab;
  const int p_total = Nb * HO * WO;
  const int p_end = min(p_begin + p_per_slab, p_total);

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  constexpr int FRB = BTB / 32;
  constexpr int FRA = BTA / 32;
  const int wm = (wid >> 1) * (BTB / 2);   // wave row (b) base
  const int wn = (wid & 1) * (BTA / 2);    // wave col (a) base

  f32x4 acc[FRB][FRA];
#pragma unroll
  for (int i = 0; i < FRB; ++i)
#pragma unroll
    for (int j = 0; j < FRA; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int SLY = (PCH * BTB / 8) / THREADS;  // Y staging slots/thread
  constexpr int SLX = (PCH * BTA / 8) / THREADS;  // X staging slots/thread
  bf16x8 yreg[SLY], xreg[SLX];

  // T14 pipeline: issue chunk t+1's global loads, MFMA chunk t from LDS,
  // write t+1 after the barrier. Pixel meta is computed inline per staging
  // slot (few int divides, hidden under the loads).
  const int HOp = HO + 2 * yring;
  const int WOp = WO + 2 * yring;
  auto load_chunk = [&](int p0) {
#pragma unroll
    for (int it = 0; it < SLY; ++it) {
      const int slot = it * THREADS + tid;
      const int pix_l = slot / (BTB / 8);
      const int ch0 = (slot % (BTB / 8)) * 8;
      const int pix = p0 + pix_l;
      bf16x8 vy = {};
      if (pix < p_end && b0 + ch0 < B) {
        long yoff = (long)pix * B;
        if (yring > 0) {
          const int n = pix / (HO * WO);
          const int rem = pix - n * (HO * WO);
          const int ho = rem / WO;
          const int wo = rem - ho * WO;
          yoff = (((long)n * HOp + ho + yring) * WOp + wo + yring) * B;
        }
        const __bf16* src = Y + yoff + b0 + ch0;
        if (b0 + ch0 + 8 <= B) {
          vy = *reinterpret_cast<const bf16x8*>(src);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (b0 + ch0 + j < B) vy[j] = src[j];
        }
      }
      yreg[it] = vy;
    }
#pragma unroll
    for (int it = 0; it < SLX; ++it) {
      const int slot = it * THREADS + tid;
      const int pix_l = slot / (BTA / 8);
      const int ch0 = (slot % (BTA / 8)) * 8;
      const int pix = p0 + pix_l;
      bf16x8 vx = {};
      if (pix < p_end) {
        const int n = pix / (HO * WO);
        const int rem = pix - n * (HO * WO);
        const int ho = rem / WO;
        const int wo = rem - ho * WO;
        const int hi = ho * STRIDE - PAD + r;
        const int wi = wo * STRIDE - PAD + s;
        if (hi >= 0 && hi < H && wi >= 0 && wi < W && a0 + ch0 < A) {
          const __bf16* src =
              X + ((long)n * H * W + (long)hi * W + wi) * A + a0 + ch0;
          if (a0 + ch0 + 8 <= A) {
            vx = *reinterpret_cast<const bf16x8*>(src);
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (a0 + ch0 + j < A) vx[j] = src[j];
          }
        }
      }
      xreg[it] = vx;
    }
  };

  auto write_chunk = [&]() {
#pragma unroll
    for (int it = 0; it < SLY; ++it) {
      const int slot = it * THREADS + tid;
      const int pix_l = slot / (BTB / 8);
      const int ch0 = (slot % (BTB / 8)) * 8;
      *reinterpret_cast<bf16x8*>(yt + sub_off<BTB>(pix_l, ch0)) = yreg[it];
    }
#pragma unroll
    for (int it = 0; it < SLX; ++it) {
      const int slot = it * THREADS + tid;
      const int pix_l = slot / (BTA / 8);
      const int ch0 = (slot % (BTA / 8)) * 8;
      *reinterpret_cast<bf16x8*>(xt + sub_off<BTA>(pix_l, ch0)) = xreg[it];
    }
  };

  const int nchunks = (p_end - p_begin + PCH - 1) / PCH;
  if (nchunks > 0) {
    load_chunk(p_begin);
    write_chunk();
    if (nchunks > 1) load_chunk(p_begin + PCH);
    __syncthreads();
  }

  for (int t = 0; t < nchunks; ++t) {
    // MFMA: 2 K-steps of 32 pixels
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int pe = kk * 32 + (lane >> 4) * 8;   // pixel base (8-aligned)
      u16x4 alo[FRB], ahi[FRB], blo[FRA], bhi[FRA];
#pragma unroll
      for (int f = 0; f < FRB; ++f)
        tr16_issue(yt, wm + f * 16 + (lane & 15), pe, lane, BTB / 16, alo[f],
                   ahi[f]);
#pragma unroll
      for (int f = 0; f < FRA; ++f)
        tr16_issue(xt, wn + f * 16 + (lane & 15), pe, lane, BTA / 16, blo[f],
                   bhi[f]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      bf16x8 a_frag[FRB], b_frag[FRA];
#pragma unroll
      for (int f = 0; f < FRB; ++f) a_frag[f] = tr16_combine(alo[f], ahi[f]);
#pragma unroll
      for (int f = 0; f < FRA; ++f) b_frag[f] = tr16_combine(blo[f], bhi[f]);
#pragma unroll
      for (int i = 0; i < FRB; ++i)
#pragma unroll
        for (int j = 0; j < FRA; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
    if (t + 1 < nchunks) {
      __syncthreads();
      write_chunk();
      if (t + 2 < nchunks) load_chunk(p_begin + (t + 2) * PCH);
      __syncthreads();
    }
  }

  // epilogue: STORE this slab's partial into its own workspace plane —
  // every (b,r,s,a) element is written by exactly one block per slab, so
  // there are no atomics and no pre-zeroing, and the serial slab walk in
  // wgrad_combine makes the weight gradient bitwise-deterministic at any
  // split factor.
  float* slab = ws + (long)blockIdx.z * B * R * S * A;
#pragma unroll
  for (int j = 0; j < FRA; ++j) {
    const int a = a0 + wn + j * 16 + (lane & 15);
    if (a >= A) continue;
#pragma unroll
    for (int i = 0; i < FRB; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const int b = b0 + wm + i * 16 + (lane >> 4) * 4 + v;
        if (b < B) {
          slab[(((long)b * R + r) * S + s) * A + a] = acc[i][j][v];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// glds-staged wgrad (round 2): for the padded-operand case (PAD==0, every X
// gather in-bounds, Y interior-mapped) staging runs on global_load_lds
// straight into the SAME [pix/4][ch/16][4][16] subtiled images the tr16
// transpose reads consume — the per-lane source address realizes the image
// permutation (16B of one pixel's 8 channels is contiguous in both). Three
// LDS buffers, counted vmcnt, raw barriers (same pipeline as
// conv2d_glds.hip); a ragged tail chunk falls back to masked register
// staging. Pixel decode is shift-only (dispatch gated on power-of-two
// interior HO/WO — every training shape qualifies).

template <int N>
__device__ __forceinline__ void wwaitcnt_vm() {
  asm volatile("s_waitcnt vmcnt(%0)" ::"n"(N) : "memory");
}

__device__ __forceinline__ void wbarrier_mem() {
  asm volatile("" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

__device__ __forceinline__ void wglds16(const __bf16* g, char* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)l, 16, 0, 0);
}

template <int BTB, int BTA>
__global__ __launch_bounds__(THREADS) void conv2d_wgrad_glds_kernel(
    const __bf16* __restrict__ Y, const __bf16* __restrict__ X,
    float* __restrict__ ws, int Nb, int HO, int WO, int B, int H, int W,
    int A, int R, int S, int STRIDE, int p_per_slab, int yring,
    int howo_sh, int wo_sh, const __bf16* __restrict__ X2, int A1) {
  extern __shared__ __align__(16) char wlds[];
  constexpr int YBYTES = PCH * BTB * 2;
  constexpr int XBYTES = PCH * BTA * 2;
  constexpr int SLAB = YBYTES + XBYTES;

  const int at_blocks = (A + BTA - 1) / BTA;
  const int b0 = (blockIdx.x / at_blocks) * BTB;
  const int a0 = (blockIdx.x % at_blocks) * BTA;
  const int r = blockIdx.y / S;
  const int s = blockIdx.y % S;
  const int p_begin = blockIdx.z * p_per_slab;
  const int p_total = Nb * HO * WO;
  const int p_end = min(p_begin + p_per_slab, p_total);

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  constexpr int FRB = BTB / 32;
  constexpr int FRA = BTA / 32;
  const int wm = (wid >> 1) * (BTB / 2);
  const int wn = (wid & 1) * (BTA / 2);
  const int HOp = HO + 2 * yring;
  const int WOp = WO + 2 * yring;

  f32x4 acc[FRB][FRA];
#pragma unroll
  for (int i = 0; i < FRB; ++i)
#pragma unroll
    for (int j = 0; j < FRA; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int YC = YBYTES / 4096;          // glds calls per wave (Y)
  constexpr int XC = XBYTES / 4096;
  constexpr int G = YC + XC;

  // per-(wave, call, lane) fixed image coordinates: 16B unit index s ->
  // subtiled-image (pix, ch0); both sides share the decode (BT via shift)
  auto decode = [&](int scall, int bt16_sh, int& pix, int& ch0) {
    const int sidx = scall >> 3;
    const int inner = scall & 7;
    pix = ((sidx >> bt16_sh) << 2) + (inner >> 1);
    ch0 = ((sidx & ((1 << bt16_sh) - 1)) << 4) + ((inner & 1) << 3);
  };
  constexpr int ybt_sh = BTB == 128 ? 3 : 2;
  constexpr int xbt_sh = BTA == 128 ? 3 : 2;

  auto stage = [&](int p0, int buf) {
    char* yimg = wlds + buf * SLAB;
    char* ximg = yimg + YBYTES;
#pragma unroll
    for (int i = 0; i < YC; ++i) {
      int pixl, ch0;
      decode((wid * YC + i) * 64 + lane, ybt_sh, pixl, ch0);
      const int pix = p0 + pixl;
      const int n = pix >> howo_sh;
      const int rem = pix - (n << howo_sh);
      const int ho = rem >> wo_sh;
      const int wo = rem - (ho << wo_sh);
      const long yoff =
          (((long)n * HOp + ho + yring) * WOp + wo + yring) * B + b0 + ch0;
      wglds16(Y + yoff, yimg + (wid * YC + i) * 1024);
    }
#pragma unroll
    for (int i = 0; i < XC; ++i) {
      int pixl, ch0;
      decode((wid * XC + i) * 64 + lane, xbt_sh, pixl, ch0);
      const int pix = p0 + pixl;
      const int n = pix >> howo_sh;
      const int rem = pix - (n << howo_sh);
      const int ho = rem >> wo_sh;
      const int wo = rem - (ho << wo_sh);
      const int hi = ho * STRIDE + r;     // PAD == 0, in-bounds by contract
      const int wi = wo * STRIDE + s;
      // dual-X (skip-concat elimination): channel picks the source tensor
      const int ach = a0 + ch0;
      const bool second = X2 != nullptr && ach >= A1;
      const __bf16* xsrc = second ? X2 : X;
      const int as = X2 == nullptr ? A : (second ? A - A1 : A1);
      const int al = second ? ach - A1 : ach;
      const long xoff = ((long)n * H * W + (long)hi * W + wi) * as + al;
      wglds16(xsrc + xoff, ximg + (wid * XC + i) * 1024);
    }
  };

  auto mfma_image = [&](const char* yimg, const char* ximg) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int pe = kk * 32 + (lane >> 4) * 8;
      u16x4 alo[FRB], ahi[FRB], blo[FRA], bhi[FRA];
#pragma unroll
      for (int f = 0; f < FRB; ++f)
        tr16_issue(yimg, wm + f * 16 + (lane & 15), pe, lane, BTB / 16,
                   alo[f], ahi[f]);
#pragma unroll
      for (int f = 0; f < FRA; ++f)
        tr16_issue(ximg, wn + f * 16 + (lane & 15), pe, lane, BTA / 16,
                   blo[f], bhi[f]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      bf16x8 a_frag[FRB], b_frag[FRA];
#pragma unroll
      for (int f = 0; f < FRB; ++f) a_frag[f] = tr16_combine(alo[f], ahi[f]);
#pragma unroll
      for (int f = 0; f < FRA; ++f) b_frag[f] = tr16_combine(blo[f], bhi[f]);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < FRB; ++i)
#pragma unroll
        for (int j = 0; j < FRA; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  const int nfull = (p_end - p_begin) / PCH;
  const int tail = (p_end - p_begin) - nfull * PCH;

  if (nfull > 0) stage(p_begin, 0);
  if (nfull > 1) stage(p_begin + PCH, 1);
  if (nfull > 2) stage(p_begin + 2 * PCH, 2);

  for (int t = 0; t < nfull; ++t) {
    const int buf = t % 3;
    // wait until chunk t's own glds completed: with fewer than 3 chunks
    // staged ahead, vmcnt(2G) would pass while chunk t is still in flight
    const int ahead = min(nfull, t + 3) - 1 - t;
    if (ahead >= 2) wwaitcnt_vm<2 * G>();
    else if (ahead == 1) wwaitcnt_vm<G>();
    else wwaitcnt_vm<0>();
    wbarrier_mem();
    mfma_image(wlds + buf * SLAB, wlds + buf * SLAB + YBYTES);
    wbarrier_mem();
    if (t + 3 < nfull) stage(p_begin + (t + 3) * PCH, buf);
  }

  if (tail > 0) {
    // ragged last chunk: masked register staging into buffer 0
    wwaitcnt_vm<0>();
    __syncthreads();
    char* yimg = wlds;
    char* ximg = wlds + YBYTES;
    const int p0 = p_begin + nfull * PCH;
    constexpr int SLY = (PCH * BTB / 8) / THREADS;
    constexpr int SLX = (PCH * BTA / 8) / THREADS;
#pragma unroll
    for (int it = 0; it < SLY; ++it) {
      const int slot = it * THREADS + tid;
      const int pixl = slot / (BTB / 8);
      const int ch0 = (slot % (BTB / 8)) * 8;
      bf16x8 vy = {};
      const int pix = p0 + pixl;
      if (pixl < tail) {
        const int n = pix >> howo_sh;
        const int rem = pix - (n << howo_sh);
        const int ho = rem >> wo_sh;
        const int wo = rem - (ho << wo_sh);
        vy = *reinterpret_cast<const bf16x8*>(
            Y + (((long)n * HOp + ho + yring) * WOp + wo + yring) * B + b0 +
            ch0);
      }
      *reinterpret_cast<bf16x8*>(yimg + sub_off<BTB>(pixl, ch0)) = vy;
    }
#pragma unroll
    for (int it = 0; it < SLX; ++it) {
      const int slot = it * THREADS + tid;
      const int pixl = slot / (BTA / 8);
      const int ch0 = (slot % (BTA / 8)) * 8;
      bf16x8 vx = {};
      const int pix = p0 + pixl;
      if (pixl < tail) {
        const int n = pix >> howo_sh;
        const int rem = pix - (n << howo_sh);
        const int ho = rem >> wo_sh;
        const int wo = rem - (ho << wo_sh);
        const int hi = ho * STRIDE + r;
        const int wi = wo * STRIDE + s;
        const int ach = a0 + ch0;
        const bool second = X2 != nullptr && ach >= A1;
        const __bf16* xsrc = second ? X2 : X;
        const int as = X2 == nullptr ? A : (second ? A - A1 : A1);
        const int al = second ? ach - A1 : ach;
        vx = *reinterpret_cast<const bf16x8*>(
            xsrc + ((long)n * H * W + (long)hi * W + wi) * as + al);
      }
      *reinterpret_cast<bf16x8*>(ximg + sub_off<BTA>(pixl, ch0)) = vx;
    }
    __syncthreads();
    mfma_image(yimg, ximg);
  }

  float* slab = ws + (long)blockIdx.z * B * R * S * A;
#pragma unroll
  for (int j = 0; j < FRA; ++j) {
    const int a = a0 + wn + j * 16 + (lane & 15);
    if (a >= A) continue;
#pragma unroll
    for (int i = 0; i < FRB; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const int b = b0 + wm + i * 16 + (lane >> 4) * 4 + v;
        if (b < B) {
          slab[(((long)b * R + r) * S + s) * A + a] = acc[i][j][v];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Small-A wgrad (the first conv: A = nc = 1 or 3 input channels). The MFMA
// kernels pay for a 64-wide A tile that is ~95% masked there — and that
// conv's wgrad was 6.6% of the whole headline step. Here the ENTIRE dW
// accumulator (B*R*S*A <= 2048 cells) lives in the block's registers
// (<= 8 cells/thread) and plain VALU rank-1 updates accumulate over an
// LDS-staged pixel tile. Slab stores + the serial combine keep it
// deterministic like the MFMA paths.

constexpr int SPIX = 32;

__global__ __launch_bounds__(THREADS) void conv2d_wgrad_smalla_kernel(
    const __bf16* __restrict__ Y, const __bf16* __restrict__ X,
    float* __restrict__ ws, int Nb, int HO, int WO, int B, int H, int W,
    int A, int R, int S, int STRIDE, int PAD, int p_per_slab, int yring,
    int howo_sh, int wo_sh) {
  __shared__ struct __align__(16) {
    float g[SPIX][64 + 1];
    float xv[SPIX][64];
  } lds;
  const int RSA = R * S * A;
  const int E = B * RSA;
  const int tid = threadIdx.x;
  const int p_begin = blockIdx.z * p_per_slab;
  const int p_total = Nb * HO * WO;
  const int p_end = min(p_begin + p_per_slab, p_total);
  const int HOp = HO + 2 * yring;
  const int WOp = WO + 2 * yring;

  // this thread's dW cells: cell = c*THREADS + tid. Decode ONCE — the
  // inner loop must be pure LDS-read + fma (a per-iteration cell/RSA
  // divide serialized the first version). Cells beyond E alias (0,0);
  // their accumulator is garbage but never stored.
  float acc[12];
  int bcell[12], jcell[12];
  const int ncells = (E + THREADS - 1) / THREADS;
#pragma unroll
  for (int c = 0; c < 12; ++c) {
    acc[c] = 0.f;
    const int cell = c * THREADS + tid;
    const int b = cell < E ? cell / RSA : 0;
    bcell[c] = b;
    jcell[c] = cell < E ? cell - b * RSA : 0;
  }

  for (int p0 = p_begin; p0 < p_end; p0 += SPIX) {
    // stage gout rows (SPIX x B) and the RSA gathers (SPIX x RSA)
    {
      const int r = tid >> 3;            // 0..31 pixel
      const int c0 = (tid & 7) * 8;      // 8 channels per thread
      const int pix = p0 + r;
      if (pix < p_end) {
        long yoff = (long)pix * B;
        if (yring > 0) {
          const int n = pix >> howo_sh;
          const int rem = pix - (n << howo_sh);
          const int ho = rem >> wo_sh;
          const int wo = rem - (ho << wo_sh);
          yoff = (((long)n * HOp + ho + yring) * WOp + wo + yring) * B;
        }
#pragma unroll
        for (int c = 0; c < 8; ++c)
          lds.g[r][c0 + c] = c0 + c < B ? (float)Y[yoff + c0 + c] : 0.f;
      } else {
#pragma unroll
        for (int c = 0; c < 8; ++c) lds.g[r][c0 + c] = 0.f;
      }
    }
    {
      // xv: each of the 256 threads stages up to ceil(SPIX*RSA/256) values
      const int nval = SPIX * RSA;
      for (int i = tid; i < SPIX * 64; i += THREADS) {
        const int r = i >> 6;            // pixel
        const int j = i & 63;            // rsa index
        float v = 0.f;
        const int pix = p0 + r;
        if (j < RSA && pix < p_end) {
          const int n = pix >> howo_sh;
          const int rem = pix - (n << howo_sh);
          const int ho = rem >> wo_sh;
          const int wo = rem - (ho << wo_sh);
          const int rs = j / A;
          const int a = j - rs * A;
          const int rr = rs / S;
          const int ss = rs - rr * S;
          const int hi = ho * STRIDE - PAD + rr;
          const int wi = wo * STRIDE - PAD + ss;
          if (hi >= 0 && hi < H && wi >= 0 && wi < W)
            v = (float)X[((long)n * H * W + (long)hi * W + wi) * A + a];
        }
        lds.xv[r][j] = v;
        (void)nval;
      }
    }
    __syncthreads();
    for (int r = 0; r < SPIX; ++r) {
#pragma unroll
      for (int c = 0; c < 12; ++c) {
        acc[c] = fmaf(lds.g[r][bcell[c]], lds.xv[r][jcell[c]], acc[c]);
      }
    }
    (void)ncells;
    __syncthreads();
  }

  float* slab = ws + (long)blockIdx.z * E;
#pragma unroll
  for (int c = 0; c < 12; ++c) {
    if (c >= ncells) break;
    const int cell = c * THREADS + tid;
    if (cell < E) slab[cell] = acc[c];
  }
}

// out[e] (+)= sum over slabs ws[z][e] — serial over z (deterministic).
__global__ __launch_bounds__(256) void wgrad_combine_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int sp, long E,
    int accumulate) {
  for (long e = (long)blockIdx.x * 256 + threadIdx.x; e < E;
       e += (long)gridDim.x * 256) {
    float t = 0.f;
    for (int z = 0; z < sp; ++z) t += ws[(long)z * E + e];
    out[e] = accumulate ? out[e] + t : t;
  }
}

}  // namespace

// Y: channels_last (N,B,HO,WO) bf16; X: channels_last (N,A,H,W) bf16.
// Returns dW (B, R, S, A) fp32. splitp<=0 -> auto. When `acc` is given it
// must be an fp32 tensor whose dense layout is (B,R,S,A) (e.g. the conv
// weight's channels_last .grad): the combine ACCUMULATES into it in place
// and returns it — this is how the framework skips autograd's per-use grad
// accumulation adds entirely (the ~29 uses per weight per step land here).
torch::Tensor conv2d_nhwc_wgrad(torch::Tensor Y, torch::Tensor X, long R,
                                long S, long stride, long pad, long splitp,
                                c10::optional<torch::Tensor> acc, long yring,
                                c10::optional<torch::Tensor> X2) {
  TORCH_CHECK(Y.is_cuda() && Y.scalar_type() == torch::kBFloat16 &&
                  Y.is_contiguous(at::MemoryFormat::ChannelsLast),
              "Y must be bf16 channels_last GPU");
  TORCH_CHECK(X.is_cuda() && X.scalar_type() == torch::kBFloat16 &&
                  X.is_contiguous(at::MemoryFormat::ChannelsLast),
              "X must be bf16 channels_last GPU");
  const int Nb = Y.size(0), B = Y.size(1);
  const int HO = Y.size(2) - 2 * (int)yring, WO = Y.size(3) - 2 * (int)yring;
  const int A1 = X.size(1);
  const int A = X2.has_value() ? A1 + (int)X2->size(1) : A1;
  const int H = X.size(2), W = X.size(3);
  if (X2.has_value()) {
    TORCH_CHECK(X2->is_cuda() && X2->scalar_type() == torch::kBFloat16 &&
                    X2->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    X2->size(0) == Nb && X2->size(2) == H &&
                    X2->size(3) == W && A1 % 64 == 0,
                "wgrad: X2 geometry mismatch");
  }
  TORCH_CHECK(X.size(0) == Nb, "batch mismatch");
  const long E = (long)B * R * S * A;

  // prefer 128-wide channel tiles; fall back to 64 when the channel count
  // is not 128-aligned (e.g. dual-X totals like 192) so the glds path's
  // exact-tiling requirement still holds
  const int BTB = (B >= 128 && B % 128 == 0) ? 128 : 64;
  const int BTA = (A >= 128 && A % 128 == 0) ? 128 : 64;
  const int bt = ceil_div(B, BTB), at = ceil_div(A, BTA);
  const int p_total = Nb * HO * WO;
  int sp = (int)splitp;
  if (sp <= 0) {
    sp = std::max(1, 1024 / std::max(1, bt * at * (int)R * (int)S));
    sp = std::min(sp, ceil_div(p_total, PCH));
  }
  const int p_per_slab =
      ceil_div(ceil_div(p_total, sp), PCH) * PCH;  // chunk-aligned
  sp = ceil_div(p_total, p_per_slab);

  // per-slab planes, written by stores — never zeroed
  auto ws = torch::empty({sp, B, (long)R, (long)S, A},
                         Y.options().dtype(torch::kFloat32));

  dim3 grid(bt * at, (int)(R * S), sp);
  auto stream = at::cuda::getCurrentCUDAStream();

  const bool pow2 = ((HO * WO) & (HO * WO - 1)) == 0 && (WO & (WO - 1)) == 0;
  const bool use_glds = pad == 0 && pow2 && HO > 0 && WO > 0 &&
                        B % BTB == 0 && A % BTA == 0 &&
                        (X2.has_value() || p_total >= 4 * PCH);
  TORCH_CHECK(!X2.has_value() || use_glds,
              "wgrad: dual-X requires the glds-eligible geometry");
  // DISABLED: three iterations (divide decode, hoisted cells, shift-only
  // decode) all measured net headline losses vs the masked MFMA tile —
  // the cost model is wrong somewhere (per-thread scalar LDS read volume
  // is the prime suspect); docs/ROADMAP.md records the attempts. Flip to
  // re-enable for tuning.
  const bool use_smalla = false && !X2.has_value() && A <= 8 && B <= 64 &&
                          R * S * A <= 64 &&
                          (long)B * R * S * A <= 3072 && pow2;
  if (use_smalla) {
    // tiny-A path: register-resident dW, VALU rank-1 updates
    const long E = (long)B * R * S * A;
    int spa = (int)std::min<long>(448, ceil_div(p_total, SPIX));
    const int pps = ceil_div(ceil_div(p_total, spa), SPIX) * SPIX;
    spa = ceil_div(p_total, pps);
    auto wsa = torch::empty({spa, B, (long)R, (long)S, A},
                            Y.options().dtype(torch::kFloat32));
    dim3 g(1, 1, spa);
    int howo_sh = 0, wo_sh = 0;
    while ((1 << howo_sh) < HO * WO) ++howo_sh;
    while ((1 << wo_sh) < WO) ++wo_sh;
    hipLaunchKernelGGL(conv2d_wgrad_smalla_kernel, g, dim3(THREADS), 0,
                       stream, reinterpret_cast<const __bf16*>(Y.data_ptr()),
                       reinterpret_cast<const __bf16*>(X.data_ptr()),
                       wsa.data_ptr<float>(), Nb, HO, WO, B, H, W, A, (int)R,
                       (int)S, (int)stride, (int)pad, pps, (int)yring,
                       howo_sh, wo_sh);
    torch::Tensor out;
    int accumulate = 0;
    if (acc.has_value()) {
      out = acc.value();
      TORCH_CHECK(out.scalar_type() == torch::kFloat32 &&
                      out.is_non_overlapping_and_dense() && out.numel() == E,
                  "wgrad acc: need dense fp32 tensor with B*R*S*A elements");
      accumulate = 1;
    } else {
      out = torch::empty({B, (long)R, (long)S, A},
                         Y.options().dtype(torch::kFloat32));
    }
    hipLaunchKernelGGL(wgrad_combine_kernel,
                       dim3((int)std::min<long>(64, (E + 255) / 256)),
                       dim3(256), 0, stream, wsa.data_ptr<float>(),
                       out.data_ptr<float>(), spa, E, accumulate);
    return out;
  }
  if (use_glds) {
    // PAD==0 means every X gather is in-bounds (padded operand or genuine
    // valid conv): glds 3-buffer pipeline, shift-only pixel decode
    int howo_sh = 0, wo_sh = 0;
    while ((1 << howo_sh) < HO * WO) ++howo_sh;
    while ((1 << wo_sh) < WO) ++wo_sh;
#define WGRAD_GLAUNCH(BB, AA)                                                 \
  do {                                                                        \
    const int shmem = 3 * PCH * (BB + AA) * 2;                                \
    static bool cfg = false;                                                  \
    if (!cfg) {                                                               \
      (void)hipFuncSetAttribute(                                              \
          (const void*)&conv2d_wgrad_glds_kernel<BB, AA>,                     \
          hipFuncAttributeMaxDynamicSharedMemorySize, shmem);                 \
      cfg = true;                                                             \
    }                                                                         \
    hipLaunchKernelGGL((conv2d_wgrad_glds_kernel<BB, AA>), grid,              \
                       dim3(THREADS), shmem, stream,                          \
                       reinterpret_cast<const __bf16*>(Y.data_ptr()),         \
                       reinterpret_cast<const __bf16*>(X.data_ptr()),         \
                       ws.data_ptr<float>(), Nb, HO, WO, B, H, W, A, (int)R,  \
                       (int)S, (int)stride, p_per_slab, (int)yring, howo_sh,  \
                       wo_sh,                                                 \
                       X2.has_value()                                         \
                           ? reinterpret_cast<const __bf16*>(X2->data_ptr())  \
                           : nullptr,                                         \
                       A1);                                                   \
  } while (0)
    if (BTB == 128 && BTA == 128) WGRAD_GLAUNCH(128, 128);
    else if (BTB == 128) WGRAD_GLAUNCH(128, 64);
    else if (BTA == 128) WGRAD_GLAUNCH(64, 128);
    else WGRAD_GLAUNCH(64, 64);
#undef WGRAD_GLAUNCH
  } else {
#define WGRAD_LAUNCH(BB, AA)                                                   \
  hipLaunchKernelGGL((conv2d_wgrad_kernel<BB, AA>), grid, dim3(THREADS), 0,    \
                     stream, reinterpret_cast<const __bf16*>(Y.data_ptr()),    \
                     reinterpret_cast<const __bf16*>(X.data_ptr()),            \
                     ws.data_ptr<float>(), Nb, HO, WO, B, H, W, A, (int)R,     \
                     (int)S, (int)stride, (int)pad, p_per_slab, (int)yring)
  if (BTB == 128 && BTA == 128) WGRAD_LAUNCH(128, 128);
  else if (BTB == 128) WGRAD_LAUNCH(128, 64);
  else if (BTA == 128) WGRAD_LAUNCH(64, 128);
  else WGRAD_LAUNCH(64, 64);
#undef WGRAD_LAUNCH
  }

  torch::Tensor out;
  int accumulate = 0;
  if (acc.has_value()) {
    out = acc.value();
    TORCH_CHECK(out.scalar_type() == torch::kFloat32 &&
                    out.is_non_overlapping_and_dense() && out.numel() == E,
                "wgrad acc: need dense fp32 tensor with B*R*S*A elements");
    accumulate = 1;
  } else {
    if (sp == 1) return ws[0];
    out = torch::empty({B, (long)R, (long)S, A},
                       Y.options().dtype(torch::kFloat32));
  }
  const int cgrid = (int)std::min<long>(2048, (E + 255) / 256);
  hipLaunchKernelGGL(wgrad_combine_kernel, dim3(cgrid), dim3(256), 0, stream,
                     ws.data_ptr<float>(), out.data_ptr<float>(), sp, E,
                     accumulate);
  return out;
}
