// Embedding gather/scatter for gfx950 — the TextCNN (IMDb config) input op.
// fwd: out[i,:] = table[ids[i],:]   (bf16, vectorized 16 B/lane)
// bwd: grad_table[ids[i],:] += dY[i,:]  (fp32 atomics into the grad arena)
// Scatter/gather is uncoalesced by nature (guide Appendix B) — rely on
// L2/L3 for the table, vectorize the row copies.

#include "lo_common.h"

namespace lo {

__global__ void embedding_fwd_kernel(const long* __restrict__ ids,
                                     const bf16* __restrict__ table,
                                     bf16* __restrict__ out,
                                     long n, int dim8) {
  const long total = n * dim8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / dim8;
    const int d = i % dim8;
    *(bf16x8*)(out + row * (long)dim8 * 8 + d * 8) =
        *(const bf16x8*)(table + ids[row] * (long)dim8 * 8 + d * 8);
  }
}

__global__ void embedding_bwd_kernel(const long* __restrict__ ids,
                                     const bf16* __restrict__ dy,
                                     float* __restrict__ gtable,
                                     long n, int dim) {
  // one WAVE per row: ids[row] is one broadcast load per row (not one per
  // element), there is no per-element index division, and each atomic issue
  // covers 64 CONTIGUOUS floats across the wave so the L2 coalesces it.
  // (A vectorized 8-elems-per-thread variant strided the wave's atomics
  // across 64 different 32 B sectors per issue — measured 9x slower.)
  const int lane = threadIdx.x & 63;
  const long wid = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const long nw = ((long)gridDim.x * blockDim.x) >> 6;
  for (long row = wid; row < n; row += nw) {
    const long t0 = ids[row] * (long)dim;
    const bf16* src = dy + row * (long)dim;
    for (int d = lane; d < dim; d += 64)
      atomicAdd(gtable + t0 + d, tofloat(src[d]));
  }
}

void launch_embedding_fwd(const void* ids, const void* table, void* out,
                          long n, int dim, hipStream_t s) {
  const int block = 256;
  const long total = n * (dim / 8);
  const int grid = (int)min((total + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(embedding_fwd_kernel, dim3(grid), dim3(block), 0, s,
                     (const long*)ids, (const bf16*)table, (bf16*)out, n, dim / 8);
}

void launch_embedding_bwd(const void* ids, const void* dy, void* gtable,
                          long n, int dim, hipStream_t s) {
  const int block = 256;                      // 4 waves -> 4 rows per block
  const int grid = (int)min((n + 3) / 4, (long)4096);
  hipLaunchKernelGGL(embedding_bwd_kernel, dim3(grid), dim3(block), 0, s,
                     (const long*)ids, (const bf16*)dy, (float*)gtable, n,
                     dim);
}

}  // namespace lo
