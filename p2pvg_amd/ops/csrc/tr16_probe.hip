// Probe for ds_read_b64_tr_b16 semantics on gfx950 (used by the GPU test
// suite to pin the lane<->element mapping before the wgrad staging uses it).
// LDS is filled with lds[i] = i (ushort); each lane supplies an address and
// the kernel dumps the 4 ushorts the instruction returned per lane.

#include "common.h"

typedef unsigned short u16x4 __attribute__((ext_vector_type(4)));

namespace {

__global__ void tr16_probe_kernel(unsigned short* __restrict__ out, int mode) {
  __shared__ unsigned short lds[1024];
  const int l = threadIdx.x & 63;
  for (int i = l; i < 1024; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  const unsigned base = (unsigned)(unsigned long long)(&lds[0]);
  unsigned off;
  switch (mode & 3) {
    case 0: off = l * 8; break;                     // each lane its own 8B
    case 1: off = (l & 15) * 8 + (l >> 4) * 128; break;
    case 2: off = (l >> 4) * 128; break;            // uniform within group
    default: off = 0; break;                        // fully uniform
  }
  const unsigned addr = base + off;
  u16x4 v;
  if (mode >= 4) {
    // control: plain ds_read_b64 at the same address
    asm volatile("ds_read_b64 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(v)
                 : "v"(addr));
  } else {
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(v)
                 : "v"(addr));
  }
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
}

}  // namespace

torch::Tensor tr16_probe(long mode) {
  auto out = torch::zeros({64, 4}, torch::TensorOptions()
                                       .dtype(torch::kInt16)
                                       .device(torch::kCUDA));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<unsigned short*>(out.data_ptr()),
                     (int)mode);
  return out;
}
