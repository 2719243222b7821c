// Elementwise / optimizer / reduction kernels for gfx950 (memory-bound class:
// guide Appendix B — always vectorize bf16 as 16 B/lane, grid-stride, cap grid).
//
// * relu_bwd          dX = dY * (Y > 0)                       (bf16)
// * sgd_step          fused momentum-SGD on the FLAT fp32 master + bf16 mirror
// * adam_step         fused Adam, same flat layout
// * colsum            dBias[n] = sum_m dY[m,n]   (bf16 -> fp32, coalesced rows)
// * argmax_rows       predictions (predict verb)
// * accuracy_count    correct-prediction counter (evaluate verb)
//
// The optimizer operates on ONE flat parameter arena (engine/params.py): the
// reference delegated optimizer steps to TF's fit; here a step
// is a single kernel over [master fp32 | grad fp32 | momentum fp32] with the
// bf16 compute mirror written in the same pass.

#include "lo_common.h"

namespace lo {

// ----------------------------------------------------------- relu_bwd ------
__global__ void relu_bwd_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ y,
                                bf16* __restrict__ dx, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 g = *(const bf16x8*)(dy + i * 8);
    bf16x8 v = *(const bf16x8*)(y + i * 8);
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = tofloat(v[j]) > 0.f ? g[j] : bf16(0.f);
    *(bf16x8*)(dx + i * 8) = o;
  }
}

void launch_relu_bwd(const void* dy, const void* y, void* dx, long n, hipStream_t s) {
  // caller guarantees n % 8 == 0 (flat activation buffers are 8-padded)
  const long n8 = n / 8;
  const int block = 256;
  const int grid = (int)min((n8 + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(relu_bwd_kernel, dim3(grid), dim3(block), 0, s,
                     (const bf16*)dy, (const bf16*)y, (bf16*)dx, n8);
}

// ----------------------------------------------------------- sgd_step ------
// master <- master - lr * (mom <- mu*mom + grad + wd*master); mirror <- bf16
__global__ void sgd_kernel(float* __restrict__ master, const float* __restrict__ grad,
                           float* __restrict__ mom, bf16* __restrict__ mirror,
                           long n4, float lr, float mu, float wd, float gscale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    f32x4 m = *(const f32x4*)(master + i * 4);
    f32x4 g = *(const f32x4*)(grad + i * 4);
    f32x4 v = *(const f32x4*)(mom + i * 4);
    bf16x4 mr;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gj = g[j] * gscale + wd * m[j];
      v[j] = mu * v[j] + gj;
      m[j] -= lr * v[j];
      mr[j] = tobf16(m[j]);
    }
    *(f32x4*)(master + i * 4) = m;
    *(f32x4*)(mom + i * 4) = v;
    *(bf16x4*)(mirror + i * 4) = mr;
  }
}

void launch_sgd(void* master, const void* grad, void* mom, void* mirror, long n,
                float lr, float mu, float wd, float gscale, hipStream_t s) {
  const long n4 = n / 4;  // arena is 8-element padded
  const int block = 256;
  const int grid = (int)min((n4 + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(sgd_kernel, dim3(grid), dim3(block), 0, s,
                     (float*)master, (const float*)grad, (float*)mom,
                     (bf16*)mirror, n4, lr, mu, wd, gscale);
}

// ---------------------------------------------------------- adam_step ------
// step_dev: device step counter (graph-capture-safe bias correction — the
// count is read on DEVICE, so replaying a captured step keeps correcting)
__global__ void adam_kernel(float* __restrict__ master, const float* __restrict__ grad,
                            float* __restrict__ m1, float* __restrict__ m2,
                            bf16* __restrict__ mirror,
                            const int* __restrict__ step_dev, long n4, float lr,
                            float b1, float b2, float eps, float wd,
                            float gscale) {
  const float t = (float)*step_dev;
  const float c1 = 1.f / (1.f - __powf(b1, t));
  const float c2 = 1.f / (1.f - __powf(b2, t));
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    f32x4 p = *(const f32x4*)(master + i * 4);
    f32x4 g = *(const f32x4*)(grad + i * 4);
    f32x4 a = *(const f32x4*)(m1 + i * 4);
    f32x4 b = *(const f32x4*)(m2 + i * 4);
    bf16x4 mr;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gj = g[j] * gscale + wd * p[j];
      a[j] = b1 * a[j] + (1.f - b1) * gj;
      b[j] = b2 * b[j] + (1.f - b2) * gj * gj;
      const float mhat = a[j] * c1;      // c1 = 1/(1-b1^t)
      const float vhat = b[j] * c2;      // c2 = 1/(1-b2^t)
      p[j] -= lr * mhat / (sqrtf(vhat) + eps);
      mr[j] = tobf16(p[j]);
    }
    *(f32x4*)(master + i * 4) = p;
    *(f32x4*)(m1 + i * 4) = a;
    *(f32x4*)(m2 + i * 4) = b;
    *(bf16x4*)(mirror + i * 4) = mr;
  }
}

void launch_adam(void* master, const void* grad, void* m1, void* m2, void* mirror,
                 const void* step_dev, long n, float lr, float b1, float b2,
                 float eps, float wd, float gscale, hipStream_t s) {
  const long n4 = n / 4;
  const int block = 256;
  const int grid = (int)min((n4 + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(adam_kernel, dim3(grid), dim3(block), 0, s,
                     (float*)master, (const float*)grad, (float*)m1, (float*)m2,
                     (bf16*)mirror, (const int*)step_dev, n4, lr, b1, b2, eps,
                     wd, gscale);
}

// ------------------------------------------------------------- colsum ------
// dBias[n] = sum_m dY[m,n] (bf16 -> fp32). Two regimes:
//  * N <= 256 (bias grads of conv/fc layers, N % 8 == 0): every thread owns a
//    FIXED 8-column chunk and strides rows — all 256 lanes stream 16 B/lane
//    coalesced; per-block LDS fp32 reduction, one global atomic per column
//    per block (guide G12). The naive column-per-thread form left 224/256
//    lanes idle at N=32 and was 30% of the MNIST step.
//  * N > 256: column-per-thread over grid.y row chunks (coalesced across N).
// mask: optional u8 array, same shape as dy; value 255 excludes the
// element (the maxpool ReLU-sentinel) — colsum(pool-level grad, idx mask)
// equals colsum of the scattered full-resolution grad, at a fraction of
// the read traffic (each pool grad value scatters at most once).
__global__ void colsum_small_kernel(const bf16* __restrict__ dy,
                                    const unsigned char* __restrict__ mask,
                                    float* __restrict__ out,
                                    long M, int N, long ldy) {
  extern __shared__ float lacc[];  // N floats
  const int nchunks = N / 8;
  const int tc = threadIdx.x % nchunks;           // column chunk (fixed)
  const int tr = threadIdx.x / nchunks;           // row lane within block
  const int rowsPerBlock = blockDim.x / nchunks;
  for (int i = threadIdx.x; i < N; i += blockDim.x) lacc[i] = 0.f;
  __syncthreads();
  float acc[8] = {};
  if (tr < rowsPerBlock) {
    const long rStride = (long)gridDim.x * rowsPerBlock;
    for (long r = (long)blockIdx.x * rowsPerBlock + tr; r < M; r += rStride) {
      bf16x8 v = *(const bf16x8*)(dy + r * ldy + tc * 8);
      if (mask) {
        const unsigned long long iv =
            *(const unsigned long long*)(mask + r * ldy + tc * 8);
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          if (((iv >> (j * 8)) & 0xffu) != 0xffu) acc[j] += tofloat(v[j]);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += tofloat(v[j]);
      }
    }
  }
  #pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(lacc + tc * 8 + j, acc[j]);
  __syncthreads();
  for (int i = threadIdx.x; i < N; i += blockDim.x) atomicAdd(out + i, lacc[i]);
}

__global__ void colsum_wide_kernel(const bf16* __restrict__ dy, float* __restrict__ out,
                                   long M, int N, long ldy, int rowsPerBlock) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  const long r0 = (long)blockIdx.y * rowsPerBlock;
  const long r1 = min(M, r0 + rowsPerBlock);
  float acc = 0.f;
  for (long r = r0; r < r1; ++r) acc += tofloat(dy[r * ldy + col]);
  if (gridDim.y == 1) out[col] = acc;
  else atomicAdd(out + col, acc);
}

void launch_colsum_masked(const void* dy, const void* mask, void* out, long M,
                          int N, long ldy, hipStream_t s);

void launch_colsum(const void* dy, void* out, long M, int N, long ldy, hipStream_t s) {
  launch_colsum_masked(dy, nullptr, out, M, N, ldy, s);
}

void launch_colsum_masked(const void* dy, const void* mask, void* out, long M,
                          int N, long ldy, hipStream_t s) {
  const int block = 256;
  if (N <= 256 && N % 8 == 0) {
    const int rowsPerBlock = block / (N / 8);
    const int grid = (int)min((M + rowsPerBlock - 1) / rowsPerBlock, (long)1024);
    hipMemsetAsync(out, 0, sizeof(float) * N, s);
    hipLaunchKernelGGL(colsum_small_kernel, dim3(grid), dim3(block),
                       N * sizeof(float), s, (const bf16*)dy,
                       (const unsigned char*)mask, (float*)out, M, N, ldy);
    return;
  }
  const int gx = (N + block - 1) / block;
  int gy = (int)min((long)(1024 / max(gx, 1) + 1), (M + 1023) / 1024);
  gy = max(gy, 1);
  const int rows = (int)((M + gy - 1) / gy);
  if (gy > 1) hipMemsetAsync(out, 0, sizeof(float) * N, s);
  hipLaunchKernelGGL(colsum_wide_kernel, dim3(gx, gy), dim3(block), 0, s,
                     (const bf16*)dy, (float*)out, M, N, ldy, rows);
}

// -------------------------------------------------------- argmax_rows ------
__global__ void argmax_kernel(const bf16* __restrict__ x, int* __restrict__ out,
                              long M, int C, int Cvalid, long ldx) {
  for (long r = (long)blockIdx.x * blockDim.x + threadIdx.x; r < M;
       r += (long)gridDim.x * blockDim.x) {
    float best = -3.0e38f;
    int bi = 0;
    for (int c = 0; c < Cvalid; ++c) {
      const float v = tofloat(x[r * ldx + c]);
      if (v > best) { best = v; bi = c; }
    }
    out[r] = bi;
  }
}

void launch_argmax_rows(const void* x, void* out, long M, int C, int Cvalid,
                        long ldx, hipStream_t s) {
  const int block = 256;
  const int grid = (int)min((M + block - 1) / block, (long)2048);
  hipLaunchKernelGGL(argmax_kernel, dim3(grid), dim3(block), 0, s,
                     (const bf16*)x, (int*)out, M, C, Cvalid, ldx);
}

// ----------------------------------------------------- accuracy_count ------
__global__ void acc_count_kernel(const int* __restrict__ pred,
                                 const long* __restrict__ label,
                                 int* __restrict__ out, long M) {
  int local = 0;
  for (long r = (long)blockIdx.x * blockDim.x + threadIdx.x; r < M;
       r += (long)gridDim.x * blockDim.x)
    local += (pred[r] == (int)label[r]);
  // wave reduce then one atomic per wave (guide G12)
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, local);
}

void launch_accuracy_count(const void* pred, const void* label, void* out, long M,
                           hipStream_t s) {
  const int block = 256;
  const int grid = (int)min((M + block - 1) / block, (long)1024);
  hipLaunchKernelGGL(acc_count_kernel, dim3(grid), dim3(block), 0, s,
                     (const int*)pred, (const long*)label, (int*)out, M);
}

}  // namespace lo
