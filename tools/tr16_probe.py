"""Empirical probe of ds_read_b64_tr_b16 lane->element mapping on gfx950."""
import torch
from torch.utils.cpp_extension import load_inline

src = r'''
#include <hip/hip_runtime.h>
#include <torch/extension.h>
typedef unsigned short u16;
typedef short s16x4 __attribute__((ext_vector_type(4)));

__global__ void probe_kernel(const u16* src, u16* out, int mode) {
  __shared__ __attribute__((aligned(16))) u16 lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = src[i];
  __syncthreads();
  const int l = threadIdx.x & 63;
  int addr_elems;
  if (mode == 0) addr_elems = l * 4;              // lane-linear 8B each
  else if (mode == 1) addr_elems = 0;             // uniform base
  else if (mode == 2) addr_elems = (l >> 4) * 64; // per-16-group 128B rows
  else addr_elems = (l & 15) * 4;                 // per-lane within group
  auto* p = (__attribute__((address_space(3))) s16x4*)(
      (__attribute__((address_space(3))) char*)lds + addr_elems * 2);
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
  #pragma unroll
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = ((u16*)&v)[j];
}

torch::Tensor probe(torch::Tensor src, int64_t mode) {
  auto out = torch::zeros({64 * 4}, src.options());
  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0,
                     (const u16*)src.data_ptr(), (u16*)out.data_ptr(), (int)mode);
  return out;
}
'''
mod = load_inline(name="trprobe",
                  cpp_sources=["torch::Tensor probe(torch::Tensor src, int64_t mode);"],
                  cuda_sources=[src],
                  functions=["probe"], with_cuda=True, verbose=False,
                  extra_cuda_cflags=["-O2"])
src_t = torch.arange(1024, dtype=torch.int16).to(torch.uint8)  # need u16...
src_t = torch.arange(1024, dtype=torch.int32).to(torch.int16).cuda()
for mode in range(4):
    out = mod.probe(src_t.view(torch.int16), mode).cpu().view(64, 4)
    print(f"mode {mode}:")
    for l in range(0, 64, 1):
        if l in (0,1,2,3,15,16,17,31,32,48,63):
            print(f"  lane {l:2d}: {[int(x) for x in out[l]]}")
