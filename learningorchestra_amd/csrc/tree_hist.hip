// Tree-histogram build for gfx950 — the hot op of the MLlib-parity tree
// family (SURVEY §2.9 "Tree histogram build + split -> per-feature histogram
// build (atomics in LDS)").
//
// hist[node][feature][bin][{grad,hess}] over samples with node_of >= 0.
// Two regimes:
//  * LDS path: a block owns a feature GROUP sized so node_group fits in LDS
//    (<=64 KiB of fp32 pairs); threads stream a row range, LDS atomics
//    accumulate, one global atomicAdd per LDS cell at flush (guide G12:
//    per-block reduction first, atomics last).
//  * global path (deep levels, many nodes): direct fp32 global atomics.

#include "lo_common.h"

namespace lo {

// fp32-pair LDS variant (reference path; also used when no scale bounds)
__global__ void tree_hist_lds_kernel(const unsigned char* __restrict__ binned,
                                     const int* __restrict__ node_of,
                                     const float* __restrict__ grad,
                                     const float* __restrict__ hess,
                                     float* __restrict__ hist,
                                     long N, int F, int f0, int FG,
                                     int n_nodes, int B) {
  extern __shared__ __attribute__((aligned(16))) float lhist[];  // [n_nodes][FG][B][2]
  const int cells = n_nodes * FG * B * 2;
  for (int i = threadIdx.x; i < cells; i += blockDim.x) lhist[i] = 0.f;
  __syncthreads();
  const long rows_per_block = (N + gridDim.y - 1) / gridDim.y;
  const long r0 = blockIdx.y * rows_per_block;
  const long r1 = min(N, r0 + rows_per_block);
  for (long r = r0 + threadIdx.x; r < r1; r += blockDim.x) {
    const int nd = node_of[r];
    if (nd < 0 || nd >= n_nodes) continue;
    const float g = grad[r], h = hess[r];
    const unsigned char* row = binned + r * F + f0;
    for (int f = 0; f < FG; ++f) {
      const int b = row[f];
      float* cell = lhist + (((nd * FG + f) * B) + b) * 2;
      atomicAdd(cell, g);
      atomicAdd(cell + 1, h);
    }
  }
  __syncthreads();
  // flush LDS -> global (hist laid out [n_nodes][F][B][2])
  for (int i = threadIdx.x; i < cells; i += blockDim.x) {
    const float v = lhist[i];
    if (v == 0.f) continue;
    const int gh = i & 1;
    const int b = (i >> 1) % B;
    const int f = (i >> 1) / B % FG;
    const int nd = (i >> 1) / B / FG;
    atomicAdd(hist + (((long)(nd * F + f0 + f) * B) + b) * 2 + gh, v);
  }
}

// u64 packed fixed-point variant: (grad, hess) quantized per LEVEL against
// host-supplied bounds and packed into ONE u64 LDS atomic per (row, feature)
// — half the LDS-atomic issue count, which is the measured wall of this
// kernel (~0.45 atomic/cycle/CU, PERFORMANCE.md r1). Field layout:
// high 32 = sum of rintf(g*Sg) (two's-complement wraps are exact as long as
// |sum| < 2^31, guaranteed by Sg = 2^30/sum|g|); low 32 = sum of
// rintf(h*Sh) with Sh = 2^31/sum(h) so the low field can never carry into
// the grad field (hess >= 0 for every loss the trees use).
__global__ void tree_hist_lds_u64_kernel(
    const unsigned char* __restrict__ binned,
    const int* __restrict__ node_of,
    const float* __restrict__ grad,
    const float* __restrict__ hess,
    float* __restrict__ hist,
    long N, int F, int f0, int FG, int n_nodes, int B,
    float sg, float sh) {
  extern __shared__ __attribute__((aligned(16))) unsigned long long lh64[];
  const int cells = n_nodes * FG * B;
  for (int i = threadIdx.x; i < cells; i += blockDim.x) lh64[i] = 0ull;
  __syncthreads();
  const long rows_per_block = (N + gridDim.y - 1) / gridDim.y;
  const long r0 = blockIdx.y * rows_per_block;
  const long r1 = min(N, r0 + rows_per_block);
  for (long r = r0 + threadIdx.x; r < r1; r += blockDim.x) {
    const int nd = node_of[r];
    if (nd < 0 || nd >= n_nodes) continue;
    const int gi = (int)rintf(grad[r] * sg);
    const unsigned int hi = (unsigned int)rintf(hess[r] * sh);
    const unsigned long long pack =
        ((unsigned long long)(unsigned int)gi << 32) | hi;
    const unsigned char* row = binned + r * F + f0;
    for (int f = 0; f < FG; ++f) {
      const int b = row[f];
      atomicAdd(lh64 + (nd * FG + f) * B + b, pack);
    }
  }
  __syncthreads();
  const float rg = 1.f / sg, rh = 1.f / sh;
  for (int i = threadIdx.x; i < cells; i += blockDim.x) {
    const unsigned long long v = lh64[i];
    if (v == 0ull) continue;
    const int b = i % B;
    const int f = i / B % FG;
    const int nd = i / B / FG;
    float* cell = hist + (((long)(nd * F + f0 + f) * B) + b) * 2;
    atomicAdd(cell, (float)(int)(v >> 32) * rg);
    atomicAdd(cell + 1, (float)(unsigned int)v * rh);
  }
}

__global__ void tree_hist_global_kernel(const unsigned char* __restrict__ binned,
                                        const int* __restrict__ node_of,
                                        const float* __restrict__ grad,
                                        const float* __restrict__ hess,
                                        float* __restrict__ hist,
                                        long N, int F, int n_nodes, int B) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < N * F;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / F;
    const int f = i % F;
    const int nd = node_of[r];
    if (nd < 0 || nd >= n_nodes) continue;
    const int b = binned[r * F + f];
    float* cell = hist + (((long)(nd * F + f) * B) + b) * 2;
    atomicAdd(cell, grad[r]);
    atomicAdd(cell + 1, hess[r]);
  }
}

void launch_tree_hist(const void* binned, const void* node_of, const void* grad,
                      const void* hess, void* hist, long N, int F, int n_nodes,
                      int B, double gbound, double hbound, hipStream_t s) {
  const int block = 256;
  // LDS budget: 64 KiB -> n_nodes*FG*B*2*4 <= 65536 => FG <= 8192/(n_nodes*B)
  const int fg_max = (int)(65536 / ((long)n_nodes * B * 2 * 4));
  if (fg_max >= 1) {
    const int FG = min(F, fg_max);
    const int fgroups = (F + FG - 1) / FG;
    int gy = (int)min((N + 4095) / 4096, (long)(1024 / fgroups + 1));
    gy = max(gy, 1);
    // packed-u64 path when the caller supplied per-level bounds (sum|g|,
    // sum h over active rows): scales keep every cell's field in range
    const bool packed = gbound > 0.0 && hbound > 0.0 &&
                        gbound < 1e30 && hbound < 1e30;
    const float sg = packed ? (float)((double)(1u << 30) / gbound) : 0.f;
    const float sh = packed ? (float)(2147483648.0 / hbound) : 0.f;
    for (int f0 = 0; f0 < F; f0 += FG) {
      const int fg = min(FG, F - f0);
      const size_t lds = (size_t)n_nodes * fg * B * 2 * 4;
      if (packed)
        hipLaunchKernelGGL(tree_hist_lds_u64_kernel, dim3(1, gy), dim3(block),
                           lds, s,
                           (const unsigned char*)binned, (const int*)node_of,
                           (const float*)grad, (const float*)hess,
                           (float*)hist, N, F, f0, fg, n_nodes, B, sg, sh);
      else
        hipLaunchKernelGGL(tree_hist_lds_kernel, dim3(1, gy), dim3(block),
                           lds, s,
                           (const unsigned char*)binned, (const int*)node_of,
                           (const float*)grad, (const float*)hess,
                           (float*)hist, N, F, f0, fg, n_nodes, B);
    }
    return;
  }
  const long total = N * F;
  const int grid = (int)min((total + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(tree_hist_global_kernel, dim3(grid), dim3(block), 0, s,
                     (const unsigned char*)binned, (const int*)node_of,
                     (const float*)grad, (const float*)hess, (float*)hist,
                     N, F, n_nodes, B);
}

}  // namespace lo
