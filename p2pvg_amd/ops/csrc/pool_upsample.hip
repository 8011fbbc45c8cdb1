// NHWC 2x2/s2 max-pool and nearest x2 upsample for gfx950
// (reference models/vgg_64.py:48 MaxPool2d(2,2),
// models/vgg_64.py:92 UpsamplingNearest2d(x2)).
//
// ATen's NHWC kernels here are scatter/atomic based; for the 2x2/s2 geometry
// every input cell belongs to exactly ONE window, so backward is a dense
// gather with no atomics. All kernels move bf16x8 vectors along C.

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef unsigned char u8x8 __attribute__((ext_vector_type(8)));

namespace {

constexpr int BLOCK = 256;

// out[n,ho,wo,c] = max of the 2x2 window; idx stores the winner (0..3)
// H/W are LOGICAL input dims; ri/ro: zero-ring widths of the physical
// input/output maps (0 = dense). i walks interior output pixels; idx stays
// dense. Padded outputs are pre-zeroed by the host (their ring is the next
// conv's padding).
__global__ __launch_bounds__(BLOCK) void maxpool2x2_fwd_kernel(
    const __bf16* __restrict__ in, __bf16* __restrict__ out,
    unsigned char* __restrict__ idx, int H, int W, int C, long nvec, int ri,
    int ro) {
  const int cvec = C / 8;
  const int WO = W / 2;
  const int Wpi = W + 2 * ri;
  const int Hpi = H + 2 * ri;
  const int WOp = WO + 2 * ro;
  const int HOp = H / 2 + 2 * ro;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    const long pix = i / cvec;          // output pixel (n*HO*WO order)
    const int c0 = (int)(i - pix * cvec) * 8;
    const long n = pix / ((long)(H / 2) * WO);
    const long rem = pix - n * ((long)(H / 2) * WO);
    const int ho = (int)(rem / WO);
    const int wo = (int)(rem - (long)ho * WO);
    const __bf16* base =
        in + ((n * Hpi + ho * 2 + ri) * Wpi + wo * 2 + ri) * C + c0;
    bf16x8 v00 = *reinterpret_cast<const bf16x8*>(base);
    bf16x8 v01 = *reinterpret_cast<const bf16x8*>(base + C);
    bf16x8 v10 = *reinterpret_cast<const bf16x8*>(base + (long)Wpi * C);
    bf16x8 v11 = *reinterpret_cast<const bf16x8*>(base + (long)Wpi * C + C);
    bf16x8 o;
    u8x8 ix;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float m = (float)v00[j];
      int k = 0;
      if ((float)v01[j] > m) { m = (float)v01[j]; k = 1; }
      if ((float)v10[j] > m) { m = (float)v10[j]; k = 2; }
      if ((float)v11[j] > m) { m = (float)v11[j]; k = 3; }
      o[j] = (__bf16)m;
      ix[j] = (unsigned char)k;
    }
    const long ooff =
        ((n * HOp + ho + ro) * WOp + wo + ro) * C + c0;
    *reinterpret_cast<bf16x8*>(out + ooff) = o;
    *reinterpret_cast<u8x8*>(idx + i * 8) = ix;
  }
}

// din[n,hi,wi,c] = gout[window] where idx selects this cell, else 0
__global__ __launch_bounds__(BLOCK) void maxpool2x2_bwd_kernel(
    const __bf16* __restrict__ gout, const unsigned char* __restrict__ idx,
    __bf16* __restrict__ din, int H, int W, int C, long nvec_in, int ri,
    int ro) {
  const int cvec = C / 8;
  const int WO = W / 2;
  const int Wpi = W + 2 * ri;
  const int Hpi = H + 2 * ri;
  const int WOp = WO + 2 * ro;
  const int HOp = H / 2 + 2 * ro;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec_in;
       i += (long)gridDim.x * BLOCK) {
    const long pix = i / cvec;          // interior input pixel
    const int c0 = (int)(i - pix * cvec) * 8;
    const long n = pix / ((long)H * W);
    const long rem = pix - n * ((long)H * W);
    const int hi = (int)(rem / W);
    const int wi = (int)(rem - (long)hi * W);
    const int k = (hi & 1) * 2 + (wi & 1);
    const long dpix = (n * (H / 2) + (hi >> 1)) * WO + (wi >> 1);  // dense
    const long opix =
        (n * HOp + (hi >> 1) + ro) * WOp + (wi >> 1) + ro;         // padded
    const bf16x8 g = *reinterpret_cast<const bf16x8*>(gout + opix * C + c0);
    const u8x8 ix = *reinterpret_cast<const u8x8*>(idx + dpix * C + c0);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (ix[j] == k) ? g[j] : (__bf16)0.f;
    const long doff = ((n * Hpi + hi + ri) * Wpi + wi + ri) * C + c0;
    *reinterpret_cast<bf16x8*>(din + doff) = o;
  }
}

// nearest x2: out[n,y,x,c] = in[n,y/2,x/2,c]
__global__ __launch_bounds__(BLOCK) void upsample2x_fwd_kernel(
    const __bf16* __restrict__ in, __bf16* __restrict__ out, int H, int W,
    int C, long nvec_out, int ri, int ro) {
  const int cvec = C / 8;
  const int WO = W * 2;
  const int Wpi = W + 2 * ri;
  const int Hpi = H + 2 * ri;
  const int WOp = WO + 2 * ro;
  const int HOp = H * 2 + 2 * ro;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec_out;
       i += (long)gridDim.x * BLOCK) {
    const long pix = i / cvec;
    const int c0 = (int)(i - pix * cvec) * 8;
    const long n = pix / ((long)H * 2 * WO);
    const long rem = pix - n * ((long)H * 2 * WO);
    const int y = (int)(rem / WO);
    const int x = (int)(rem - (long)y * WO);
    const bf16x8 v = *reinterpret_cast<const bf16x8*>(
        in + ((n * Hpi + (y >> 1) + ri) * Wpi + (x >> 1) + ri) * C + c0);
    const long ooff = ((n * HOp + y + ro) * WOp + x + ro) * C + c0;
    *reinterpret_cast<bf16x8*>(out + ooff) = v;
  }
}

// bwd: din[n,hi,wi,c] = sum of the 4 output positions that sampled it
__global__ __launch_bounds__(BLOCK) void upsample2x_bwd_kernel(
    const __bf16* __restrict__ gout, __bf16* __restrict__ din, int H, int W,
    int C, long nvec_in, int ri, int ro) {
  const int cvec = C / 8;
  const int WO = W * 2;
  const int Wpi = W + 2 * ri;
  const int Hpi = H + 2 * ri;
  const int WOp = WO + 2 * ro;
  const int HOp = H * 2 + 2 * ro;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec_in;
       i += (long)gridDim.x * BLOCK) {
    const long pix = i / cvec;
    const int c0 = (int)(i - pix * cvec) * 8;
    const long n = pix / ((long)H * W);
    const long rem = pix - n * ((long)H * W);
    const int hi = (int)(rem / W);
    const int wi = (int)(rem - (long)hi * W);
    const __bf16* base =
        gout + ((n * HOp + hi * 2 + ro) * WOp + wi * 2 + ro) * C + c0;
    bf16x8 a = *reinterpret_cast<const bf16x8*>(base);
    bf16x8 b = *reinterpret_cast<const bf16x8*>(base + C);
    bf16x8 c = *reinterpret_cast<const bf16x8*>(base + (long)WOp * C);
    bf16x8 d = *reinterpret_cast<const bf16x8*>(base + (long)WOp * C + C);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (__bf16)((float)a[j] + (float)b[j] + (float)c[j] + (float)d[j]);
    const long doff = ((n * Hpi + hi + ri) * Wpi + wi + ri) * C + c0;
    *reinterpret_cast<bf16x8*>(din + doff) = o;
  }
}

int grid_for(long nvec) {
  return (int)std::min<long>(2048, (nvec + BLOCK - 1) / BLOCK);
}

void check_in(const torch::Tensor& t) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
                  t.is_contiguous(at::MemoryFormat::ChannelsLast),
              "expected bf16 channels_last GPU tensor");
  TORCH_CHECK(t.size(1) % 8 == 0, "C must be a multiple of 8");
}

}  // namespace

// ri: zero-ring width of the (physical) input map; ro: ring of the output.
// Padded outputs are allocated ZEROED (their ring is the consuming conv's
// padding); gradient outputs are allocated empty (their ring is garbage the
// interior-reading consumers ignore).
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor in, long ri, long ro) {
  check_in(in);
  const int N = in.size(0), C = in.size(1);
  const int H = in.size(2) - 2 * (int)ri, W = in.size(3) - 2 * (int)ri;
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0 && H > 0, "even spatial dims only");
  auto opts = in.options().memory_format(at::MemoryFormat::ChannelsLast);
  auto out = ro > 0
                 ? torch::zeros({N, C, H / 2 + 2 * ro, W / 2 + 2 * ro}, opts)
                 : torch::empty({N, C, H / 2, W / 2}, opts);
  auto idx = torch::empty({(long)N * (H / 2) * (W / 2) * C},
                          in.options().dtype(torch::kByte));
  const long nvec = (long)N * (H / 2) * (W / 2) * C / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(maxpool2x2_fwd_kernel, dim3(grid_for(nvec)), dim3(BLOCK),
                     0, stream, reinterpret_cast<const __bf16*>(in.data_ptr()),
                     reinterpret_cast<__bf16*>(out.data_ptr()),
                     idx.data_ptr<unsigned char>(), H, W, C, nvec, (int)ri,
                     (int)ro);
  return {out, idx};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor gout, torch::Tensor idx, long H,
                             long W, long ri, long ro) {
  check_in(gout);
  const int N = gout.size(0), C = gout.size(1);
  auto din = torch::empty({N, C, H + 2 * ri, W + 2 * ri},
                          gout.options().memory_format(at::MemoryFormat::ChannelsLast));
  const long nvec = (long)N * H * W * C / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(maxpool2x2_bwd_kernel, dim3(grid_for(nvec)), dim3(BLOCK),
                     0, stream,
                     reinterpret_cast<const __bf16*>(gout.data_ptr()),
                     idx.data_ptr<unsigned char>(),
                     reinterpret_cast<__bf16*>(din.data_ptr()), (int)H, (int)W,
                     C, nvec, (int)ri, (int)ro);
  return din;
}

torch::Tensor upsample2x_fwd(torch::Tensor in, long ri, long ro) {
  check_in(in);
  const int N = in.size(0), C = in.size(1);
  const int H = in.size(2) - 2 * (int)ri, W = in.size(3) - 2 * (int)ri;
  auto opts = in.options().memory_format(at::MemoryFormat::ChannelsLast);
  auto out = ro > 0
                 ? torch::zeros({N, C, H * 2 + 2 * ro, W * 2 + 2 * ro}, opts)
                 : torch::empty({N, C, H * 2, W * 2}, opts);
  const long nvec = (long)N * H * 2 * W * 2 * C / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(upsample2x_fwd_kernel, dim3(grid_for(nvec)), dim3(BLOCK),
                     0, stream, reinterpret_cast<const __bf16*>(in.data_ptr()),
                     reinterpret_cast<__bf16*>(out.data_ptr()), H, W, C, nvec,
                     (int)ri, (int)ro);
  return out;
}

torch::Tensor upsample2x_bwd(torch::Tensor gout, long ri, long ro) {
  check_in(gout);
  const int N = gout.size(0), C = gout.size(1);
  const int H = (gout.size(2) - 2 * (int)ro) / 2,
            W = (gout.size(3) - 2 * (int)ro) / 2;
  auto din = torch::empty({N, C, H + 2 * ri, W + 2 * ri},
                          gout.options().memory_format(at::MemoryFormat::ChannelsLast));
  const long nvec = (long)N * H * W * C / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(upsample2x_bwd_kernel, dim3(grid_for(nvec)), dim3(BLOCK),
                     0, stream,
                     reinterpret_cast<const __bf16*>(gout.data_ptr()),
                     reinterpret_cast<__bf16*>(din.data_ptr()), H, W, C, nvec,
                     (int)ri, (int)ro);
  return din;
}
