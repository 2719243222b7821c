// BatchNorm (NHWC, per-channel) + residual add + global average pool for
// gfx950 — the ResNet-50 block set (SURVEY §2.9 "BatchNorm / activation
// fusions -> fused BN+ReLU").
//
// Layout: x viewed as [M, C] with M = B*H*W, C % 8 == 0. All reductions use
// the colsum_small pattern (fixed 8-column chunk per thread, full-row
// coalesced streams, LDS reduce, one global atomic per cell per block).
//
//  bn_stats      x -> sum[c], sumsq[c]
//  bn_fwd        y = gamma*(x-mean)*invstd + beta (+ReLU), elementwise
//  bn_bwd_reduce dbeta[c] = sum dy', dgamma[c] = sum dy'*xhat
//                (dy' = dy masked by y>0 when ReLU was fused)
//  bn_bwd_dx     dx = gamma*invstd*(dy' - dbeta/M - xhat*dgamma/M)
//  add_relu      z = relu(a + b) (residual join; backward = relu_bwd)
//  avgpool_gl    [B,HW,C] -> [B,C] mean; bwd broadcast

#include "lo_common.h"

namespace lo {

// generic dual-column reduction: each thread owns an 8-col chunk; works for
// any C % 8 == 0 (C/8 <= blockDim keeps every lane busy; larger C loops).
// accum(r, c0, acc1[8], acc2[8]): vector-load the 8-column chunk at (r, c0)
// and add into the accumulators.
template <typename ACCUM>
__device__ __forceinline__ void colreduce2(long M, int C, int ldx,
                                           float* out1, float* out2,
                                           ACCUM accum) {
  extern __shared__ __attribute__((aligned(16))) float lacc[];  // [2][C]
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lacc[i] = 0.f;
  __syncthreads();
  const int nchunks = C / 8;
  float acc1[8] = {}, acc2[8] = {};
  const int chunksPerBlock = min(nchunks, (int)blockDim.x);
  // thread layout: tc = chunk, tr = row lane; loop outer over chunk phases
  for (int phase = 0; phase < (nchunks + chunksPerBlock - 1) / chunksPerBlock;
       ++phase) {
    const int tc = phase * chunksPerBlock + (threadIdx.x % chunksPerBlock);
    const int tr = threadIdx.x / chunksPerBlock;
    const int rowsPerBlock = blockDim.x / chunksPerBlock;
    if (tc < nchunks && tr < rowsPerBlock) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) { acc1[j] = 0.f; acc2[j] = 0.f; }
      const long rStride = (long)gridDim.x * rowsPerBlock;
      for (long r = (long)blockIdx.x * rowsPerBlock + tr; r < M; r += rStride)
        accum(r, tc * 8, acc1, acc2);
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(lacc + tc * 8 + j, acc1[j]);
        atomicAdd(lacc + C + tc * 8 + j, acc2[j]);
      }
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < C; i += blockDim.x) {
    atomicAdd(out1 + i, lacc[i]);
    atomicAdd(out2 + i, lacc[C + i]);
  }
}

__global__ void bn_stats_kernel(const bf16* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long M, int C) {
  colreduce2(M, C, C, sum, sumsq,
             [&](long r, int c0, float (&a1)[8], float (&a2)[8]) {
               bf16x8 v = *(const bf16x8*)(x + r * C + c0);
               #pragma unroll
               for (int j = 0; j < 8; ++j) {
                 const float f = tofloat(v[j]);
                 a1[j] += f;
                 a2[j] += f * f;
               }
             });
}

// fixed-chunk thread layout: tc/tr computed once, per-channel params hoisted
// out of the row loop (the flat-index form was VALU-bound on 64-bit div/mod
// and re-loaded 4-5 scalar params per element — 18% of the ResNet step).
// residual: optional second input added before the (optional) ReLU — the
// ResNet bottleneck join z = relu(bn(x) + idt) fused into the BN pass.
__global__ void bn_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              const bf16* __restrict__ residual,
                              long M, int C, int relu) {
  const int nch = C / 8;
  for (int phase = 0; phase * (int)blockDim.x < nch; ++phase) {
    const int chunksPerBlock = min(nch - phase * (int)blockDim.x, (int)blockDim.x);
    const int tc = phase * blockDim.x + threadIdx.x % chunksPerBlock;
    const int tr = threadIdx.x / chunksPerBlock;
    const int rowsPerBlock = blockDim.x / chunksPerBlock;
    if (tr >= rowsPerBlock) continue;
    const int c0 = tc * 8;
    float mn[8], is[8], gm[8], bt[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      mn[j] = mean[c0 + j]; is[j] = invstd[c0 + j];
      gm[j] = gamma[c0 + j]; bt[j] = beta[c0 + j];
    }
    const long rStride = (long)gridDim.x * rowsPerBlock;
    for (long r = (long)blockIdx.x * rowsPerBlock + tr; r < M; r += rStride) {
      bf16x8 v = *(const bf16x8*)(x + r * C + c0);
      bf16x8 res;
      if (residual) res = *(const bf16x8*)(residual + r * C + c0);
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (tofloat(v[j]) - mn[j]) * is[j] * gm[j] + bt[j];
        if (residual) f += tofloat(res[j]);
        if (relu) f = fmaxf(f, 0.f);
        o[j] = tobf16(f);
      }
      *(bf16x8*)(y + r * C + c0) = o;
    }
  }
}

// dy' = dy (masked by y>0 if relu); dbeta = sum dy', dgamma = sum dy'*xhat
__global__ void bn_bwd_reduce_kernel(const bf16* __restrict__ dy,
                                     const bf16* __restrict__ y,
                                     const bf16* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ dbeta,
                                     float* __restrict__ dgamma,
                                     long M, int C, int relu) {
  colreduce2(M, C, C, dbeta, dgamma,
             [&](long r, int c0, float (&a1)[8], float (&a2)[8]) {
               bf16x8 gv = *(const bf16x8*)(dy + r * C + c0);
               bf16x8 xv = *(const bf16x8*)(x + r * C + c0);
               bf16x8 yv;
               if (relu) yv = *(const bf16x8*)(y + r * C + c0);
               #pragma unroll
               for (int j = 0; j < 8; ++j) {
                 float g = tofloat(gv[j]);
                 if (relu && tofloat(yv[j]) <= 0.f) g = 0.f;
                 const int c = c0 + j;
                 a1[j] += g;
                 a2[j] += g * (tofloat(xv[j]) - mean[c]) * invstd[c];
               }
             });
}

__global__ void bn_bwd_dx_kernel(const bf16* __restrict__ dy,
                                 const bf16* __restrict__ y,
                                 const bf16* __restrict__ x,
                                 bf16* __restrict__ dx,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ dbeta,
                                 const float* __restrict__ dgamma,
                                 long M, int C, int relu) {
  const float invM = 1.f / (float)M;
  const int nch = C / 8;
  for (int phase = 0; phase * (int)blockDim.x < nch; ++phase) {
    const int chunksPerBlock = min(nch - phase * (int)blockDim.x, (int)blockDim.x);
    const int tc = phase * blockDim.x + threadIdx.x % chunksPerBlock;
    const int tr = threadIdx.x / chunksPerBlock;
    const int rowsPerBlock = blockDim.x / chunksPerBlock;
    if (tr >= rowsPerBlock) continue;
    const int c0 = tc * 8;
    // per-channel factors folded once: d = A*g + Bx*xhat + Cc
    float A_[8], Bx[8], Cc[8], mn[8], is[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      const float gi = gamma[c] * invstd[c];
      A_[j] = gi;
      Bx[j] = -gi * dgamma[c] * invM;
      Cc[j] = -gi * dbeta[c] * invM;
      mn[j] = mean[c];
      is[j] = invstd[c];
    }
    const long rStride = (long)gridDim.x * rowsPerBlock;
    for (long r = (long)blockIdx.x * rowsPerBlock + tr; r < M; r += rStride) {
      bf16x8 gv = *(const bf16x8*)(dy + r * C + c0);
      bf16x8 xv = *(const bf16x8*)(x + r * C + c0);
      bf16x8 yv;
      if (relu) yv = *(const bf16x8*)(y + r * C + c0);
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = tofloat(gv[j]);
        if (relu && tofloat(yv[j]) <= 0.f) g = 0.f;
        const float xhat = (tofloat(xv[j]) - mn[j]) * is[j];
        o[j] = tobf16(A_[j] * g + Bx[j] * xhat + Cc[j]);
      }
      *(bf16x8*)(dx + r * C + c0) = o;
    }
  }
}

// fused stats finalization: mean/invstd from the (sum, sumsq) scratch AND
// the running-stats EMA in ONE kernel — replaces the ~12 tiny torch
// elementwise launches per BN layer per step (~3% of the ResNet step in
// launch overhead, r2 profile). Capture-safe (pure device arithmetic).
__global__ void bn_finalize_stats_kernel(const float* __restrict__ scratch,
                                         float* __restrict__ mean,
                                         float* __restrict__ invstd,
                                         float* __restrict__ rmean,
                                         float* __restrict__ rvar,
                                         float invM, float m, float eps,
                                         int C) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < C) {
    const float mu = scratch[i] * invM;
    float var = scratch[C + i] * invM - mu * mu;
    var = var > 0.f ? var : 0.f;
    mean[i] = mu;
    invstd[i] = rsqrtf(var + eps);
    if (rmean) {
      rmean[i] += (mu - rmean[i]) * m;
      rvar[i] += (var - rvar[i]) * m;
    }
  }
}

__global__ void add_relu_kernel(const bf16* __restrict__ a,
                                const bf16* __restrict__ b,
                                bf16* __restrict__ z, long n8, int relu) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 va = *(const bf16x8*)(a + i * 8);
    bf16x8 vb = *(const bf16x8*)(b + i * 8);
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = tofloat(va[j]) + tofloat(vb[j]);
      if (relu) f = fmaxf(f, 0.f);
      o[j] = tobf16(f);
    }
    *(bf16x8*)(z + i * 8) = o;
  }
}

// global average pool: x [B, HW, C] -> out [B, C]; one block per image row
// group (C % 8 == 0)
__global__ void avgpool_gl_fwd_kernel(const bf16* __restrict__ x,
                                      bf16* __restrict__ out,
                                      int B, int HW, int C) {
  const int b = blockIdx.x;
  const float inv = 1.f / (float)HW;
  for (int c0 = threadIdx.x * 8; c0 < C; c0 += blockDim.x * 8) {
    float acc[8] = {};
    for (int p = 0; p < HW; ++p) {
      bf16x8 v = *(const bf16x8*)(x + ((long)b * HW + p) * C + c0);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += tofloat(v[j]);
    }
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = tobf16(acc[j] * inv);
    *(bf16x8*)(out + (long)b * C + c0) = o;
  }
}

__global__ void avgpool_gl_bwd_kernel(const bf16* __restrict__ dy,
                                      bf16* __restrict__ dx,
                                      int B, int HW, int C) {
  const long total = (long)B * HW * (C / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int c0 = (int)(i % (C / 8)) * 8;
    const long bp = i / (C / 8);
    const int b = (int)(bp / HW);
    bf16x8 v = *(const bf16x8*)(dy + (long)b * C + c0);
    bf16x8 o;
    const float inv = 1.f / (float)HW;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = tobf16(tofloat(v[j]) * inv);
    *(bf16x8*)(dx + bp * C + c0) = o;
  }
}

// ----------------------------------------------------------- launchers ----
static int _grid(long work, int block) {
  return (int)min((work + block - 1) / block, (long)2048);
}

void launch_bn_stats(const void* x, void* sum, void* sumsq, long M, int C,
                     hipStream_t s) {
  hipMemsetAsync(sum, 0, 4 * C, s);
  hipMemsetAsync(sumsq, 0, 4 * C, s);
  const int block = 256;
  const int rowsPerBlock = max(1, block / (C / 8));
  const int grid = (int)min((M + rowsPerBlock - 1) / rowsPerBlock, (long)1024);
  hipLaunchKernelGGL(bn_stats_kernel, dim3(grid), dim3(block), 2 * C * 4, s,
                     (const bf16*)x, (float*)sum, (float*)sumsq, M, C);
}

void launch_bn_fwd(const void* x, void* y, const void* mean, const void* invstd,
                   const void* gamma, const void* beta, const void* residual,
                   long M, int C, int relu, hipStream_t s) {
  const int block = 256;
  const int rowsPerBlock = max(1, block / (C / 8));
  const int grid = (int)min((M + rowsPerBlock - 1) / rowsPerBlock, (long)2048);
  hipLaunchKernelGGL(bn_fwd_kernel, dim3(grid), dim3(block),
                     0, s, (const bf16*)x, (bf16*)y, (const float*)mean,
                     (const float*)invstd, (const float*)gamma,
                     (const float*)beta, (const bf16*)residual, M, C, relu);
}

void launch_bn_bwd_reduce(const void* dy, const void* y, const void* x,
                          const void* mean, const void* invstd, void* dbeta,
                          void* dgamma, long M, int C, int relu, hipStream_t s) {
  hipMemsetAsync(dbeta, 0, 4 * C, s);
  hipMemsetAsync(dgamma, 0, 4 * C, s);
  const int block = 256;
  const int rowsPerBlock = max(1, block / (C / 8));
  const int grid = (int)min((M + rowsPerBlock - 1) / rowsPerBlock, (long)1024);
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(grid), dim3(block), 2 * C * 4, s,
                     (const bf16*)dy, (const bf16*)y, (const bf16*)x,
                     (const float*)mean, (const float*)invstd, (float*)dbeta,
                     (float*)dgamma, M, C, relu);
}

void launch_bn_bwd_dx(const void* dy, const void* y, const void* x, void* dx,
                      const void* mean, const void* invstd, const void* gamma,
                      const void* dbeta, const void* dgamma, long M, int C,
                      int relu, hipStream_t s) {
  const int block = 256;
  const int rowsPerBlock = max(1, block / (C / 8));
  const int grid = (int)min((M + rowsPerBlock - 1) / rowsPerBlock, (long)2048);
  hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(grid),
                     dim3(block), 0, s, (const bf16*)dy, (const bf16*)y,
                     (const bf16*)x, (bf16*)dx, (const float*)mean,
                     (const float*)invstd, (const float*)gamma,
                     (const float*)dbeta, (const float*)dgamma, M, C, relu);
}

void launch_bn_finalize_stats(const void* scratch, void* mean, void* invstd,
                              void* rmean, void* rvar, float invM,
                              float momentum, float eps, int C,
                              hipStream_t s) {
  hipLaunchKernelGGL(bn_finalize_stats_kernel, dim3((C + 255) / 256),
                     dim3(256), 0, s, (const float*)scratch, (float*)mean,
                     (float*)invstd, (float*)rmean, (float*)rvar, invM,
                     momentum, eps, C);
}

void launch_add_relu(const void* a, const void* b, void* z, long n, int relu,
                     hipStream_t s) {
  const int block = 256;
  hipLaunchKernelGGL(add_relu_kernel, dim3(_grid(n / 8, block)), dim3(block),
                     0, s, (const bf16*)a, (const bf16*)b, (bf16*)z, n / 8, relu);
}

void launch_avgpool_global(const void* x, void* out, int B, int HW, int C,
                           hipStream_t s) {
  hipLaunchKernelGGL(avgpool_gl_fwd_kernel, dim3(B), dim3(256), 0, s,
                     (const bf16*)x, (bf16*)out, B, HW, C);
}

void launch_avgpool_global_bwd(const void* dy, void* dx, int B, int HW, int C,
                               hipStream_t s) {
  const int block = 256;
  hipLaunchKernelGGL(avgpool_gl_bwd_kernel,
                     dim3(_grid((long)B * HW * (C / 8), block)), dim3(block),
                     0, s, (const bf16*)dy, (bf16*)dx, B, HW, C);
}

}  // namespace lo
