// Ablation probe for conv_dx_kernel: which phase eats the time?
// Standalone hipcc binary (no torch). ABL: 0=full, 1=no scatter, 2=no mfma,
// 3=no barriers (single buffer, racy - timing only), 4=scatter-only.
#include <hip/hip_runtime.h>
#include <cstdio>
#include "../learningorchestra_amd/csrc/lo_common.h"
using namespace lo;

template <int ABL>
__global__ __launch_bounds__(256) void dxk(
    const bf16* __restrict__ dy2, long ldy, const bf16* __restrict__ wt,
    long ldw, bf16* __restrict__ dx, int H, int W, int C, int KH, int KW,
    int SH, int SW, int PH, int PW, int OH, int OW, int outC) {
  extern __shared__ float ldx[];
  const int img = blockIdx.x;
  const int R = OH * OW;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int HWC = H * W * C;
  char* smW = (char*)(ldx + HWC);
  for (int i = tid; i < HWC; i += 256) ldx[i] = 0.f;

  const bf16* dyi = dy2 + (long)img * OW * OH * ldy;
  const int CB = C > 32 ? 32 : C;
  const int NSTEP = KH * KW * (C / CB);
  const int srow = tid >> 3, skc = tid & 7;
  auto tile_n0 = [&](int step) {
    const int cs = C / CB;
    return (step / cs) * C + (step % cs) * CB;
  };
  auto fetch_tile = [&](int step) -> bf16x8 {
    bf16x8 v = {};
    if (step < NSTEP && srow < CB && skc * 8 < outC)
      v = *(const bf16x8*)(wt + (long)(tile_n0(step) + srow) * ldw + skc * 8);
    return v;
  };
  auto write_tile = [&](int buf, bf16x8 v) {
    *(bf16x8*)(smW + buf * 4096 + srow * 128 + ((skc * 16) ^ ((srow & 7) << 4))) = v;
  };

  for (int m0 = 0; m0 < R; m0 += 64) {
    const int arow = m0 + wave * 16 + (lane & 15);
    bf16x8 af[2] = {};
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int k = kc * 32 + (lane >> 4) * 8;
      if (arow < R && k < outC)
        af[kc] = *(const bf16x8*)(dyi + (long)arow * ldy + k);
    }
    int oh4[4], ow4[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      oh4[r] = m / OW;
      ow4[r] = m - oh4[r] * OW;
    }
    const int mok = (m0 + wave * 16 + (lane >> 4) * 4 + 3 < R) ? 4
                    : max(0, R - (m0 + wave * 16 + (lane >> 4) * 4));

    bf16x8 stg = fetch_tile(0);
    write_tile(0, stg);
    for (int step = 0; step < NSTEP; ++step) {
      const int buf = (ABL == 3) ? 0 : (step & 1);
      stg = fetch_tile(step + 1);
      if (ABL != 3) __syncthreads();
      const int cs = C / CB;
      const int khkw = step / cs, cb = (step % cs) * CB;
      const int kh = khkw / KW, kw = khkw - kh * KW;
      f32x4 acc[2] = {};
      if (ABL != 2) {
        #pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int k = kc * 32 + (lane >> 4) * 8;
          #pragma unroll
          for (int ni = 0; ni < 2; ++ni) {
            const int row = ni * 16 + (lane & 15);
            const bf16x8 bfr = *(const bf16x8*)(
                smW + buf * 4096 + row * 128 + ((k * 2) ^ ((row & 7) << 4)));
            acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[kc], bfr, acc[ni], 0, 0, 0);
          }
        }
      } else {
        acc[0][0] = tofloat(stg[0]);  // keep a data dependency
      }
      if (ABL != 1 && ABL != 4) {
        const int hb = -PH + kh, wb = -PW + kw;
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          if (ni * 16 >= CB) break;
          const int c = cb + ni * 16 + (lane & 15);
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            if (r >= mok) break;
            const int h = oh4[r] * SH + hb;
            const int w = ow4[r] * SW + wb;
            if (h >= 0 && h < H && w >= 0 && w < W)
              atomicAdd(ldx + (h * W + w) * C + c, acc[ni][r]);
          }
        }
      } else if (ABL == 4) {
        const int hb = -PH + kh, wb = -PW + kw;
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const int c = cb + ni * 16 + (lane & 15);
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int h = oh4[r] * SH + hb;
            const int w = ow4[r] * SW + wb;
            if (h >= 0 && h < H && w >= 0 && w < W)
              atomicAdd(ldx + (h * W + w) * C + c, tofloat(stg[r]));
          }
        }
      }
      if (ABL != 3) __syncthreads();
      write_tile(buf ^ 1, stg);
    }
  }
  __syncthreads();
  bf16* dxi = dx + (long)img * HWC;
  for (int i = tid * 8; i < HWC; i += 256 * 8) {
    bf16x8 v;
    #pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tobf16(ldx[i + j]);
    *(bf16x8*)(dxi + i) = v;
  }
}

int main() {
  const int B = 32768, H = 12, W = 12, C = 32, KH = 5, outC = 64;
  const int OH = H - KH + 1, OW = OH, R = OH * OW;
  const int kpad = KH * KH * C, HWC = H * W * C;
  bf16 *dy2, *wt, *dx;
  (void)hipMalloc(&dy2, (long)B * R * outC * 2);
  (void)hipMalloc(&wt, (long)kpad * outC * 2);
  (void)hipMalloc(&dx, (long)B * HWC * 2);
  (void)hipMemset(dy2, 0x3c, (long)B * R * outC * 2);
  (void)hipMemset(wt, 0x3c, (long)kpad * outC * 2);
  const int lds = HWC * 4 + 8192;
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  auto run = [&](auto kern, const char* name) {
    for (int i = 0; i < 3; ++i)
      hipLaunchKernelGGL(kern, dim3(B), dim3(256), lds, 0, dy2, (long)outC,
                         wt, (long)outC, dx, H, W, C, KH, KH, 1, 1, 0, 0, OH,
                         OW, outC);
    (void)hipEventRecord(e0);
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL(kern, dim3(B), dim3(256), lds, 0, dy2, (long)outC,
                         wt, (long)outC, dx, H, W, C, KH, KH, 1, 1, 0, 0, OH,
                         OW, outC);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    printf("%-12s %8.3f ms\n", name, ms / 10);
  };
  run(dxk<0>, "full");
  run(dxk<1>, "no-scatter");
  run(dxk<2>, "no-mfma");
  run(dxk<3>, "no-barrier");
  run(dxk<4>, "scatter-only");
  (void)hipDeviceSynchronize();
  printf("err=%d\n", (int)hipGetLastError());
  return 0;
}
