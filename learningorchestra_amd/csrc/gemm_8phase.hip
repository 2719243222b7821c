// 256x256 8-phase MFMA GEMM for gfx950 — the deep-pipelined large-shape path
// (guide §5 "The 256² 8-phase template"): 8 waves (2Mx4N), BK=64, two 64-KiB
// LDS buffers, one half-tile staged per phase by global_load_lds, COUNTED
// s_waitcnt vmcnt(4) once per K-tile (never a full drain in the main loop),
// raw s_barrier pairs around each 16-MFMA phase, s_setprio around the MFMA
// cluster (guide T5).
//
// The staging schedule was derived with an explicit region-lifetime + vmcnt
// FIFO simulator (tools/sched_sim.py; 156 valid schedules, this is the
// latest-staging one):
//   tile s computes in phases q0..q3 (globally 4s+1..4s+4), buffer s%2
//   q0: read B frags (8 ds_read_b128) + A quadrant 0
//   q1: A quadrant 1 reads; STAGE A0(s+1)
//   q2: A quadrant 2 reads; STAGE A1(s+1)
//   q3: A quadrant 3 reads; STAGE B0(s+2)+B1(s+2); s_waitcnt vmcnt(4)
// Overwrite safety: a region's staging glds issues only in a phase after the
// barrier that closes its previous content's last read (B halves are fully
// consumed in q0, A halves by q3). Read safety: the q3 vmcnt(4) leaves only
// the 4 newest glds (the two B-half stages of that phase) outstanding, so
// every half of tile s+1 is complete before its q0.
//
// Covers (F,T) bf16-out shapes with M%256==0, N%256==0, K%64==0, K>=256.
// Same LDS row-major images + XOR(row&7)<<4 swizzle as gemm.hip (glds
// source-side swizzle, rule 21).

#include "lo_common.h"

namespace lo {

template <int EPI, bool DRAIN>
__global__ __launch_bounds__(512) void gemm256_kernel(
    const bf16* __restrict__ A, long lda, const bf16* __restrict__ B, long ldb,
    bf16* __restrict__ C, long ldc, const float* __restrict__ bias,
    int M, int N, int K, const bf16* __restrict__ addend,
    float* __restrict__ stats_sum, float* __restrict__ stats_sumsq) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int BKB = BK * 2;               // 128-B LDS rows
  constexpr int MFRAG = 8, NFRAG = 4;       // per-wave 128x64 output
  constexpr int BUFB = (BM + BN) * BKB;     // 64 KiB per buffer

  extern __shared__ __attribute__((aligned(16))) char smem[];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;                 // 8 waves
  const int wr = wid >> 2, wc = wid & 3;    // 2 x 4

  const int nb = N / BN;
  const int mb = M / BM;
  int bid = xcd_swizzle(blockIdx.x, mb * nb);
  const int bm = bid % mb, bn = bid / mb;
  const int m0 = bm * BM, n0 = bn * BN;

  const int fr = lane & 15;
  const int fkb = (lane >> 4) * 8;
  const int ntiles = K / BK;

  f32x4 acc[MFRAG][NFRAG] = {};
  bf16x8 bfr[NFRAG][2];                     // B frags held across a tile

  // ---- staging: one half-tile (128 rows x 64 k = 16 KiB) by 8 waves, two
  // 1-KiB glds segments per wave (16 segments). half: 0 = rows 0-127 of the
  // A image, 1 = rows 128-255; likewise for B at image offset BM*BKB.
  auto stage_half = [&](int buf, int k0, int imgOff, int half,
                        const bf16* __restrict__ G, long ldg, int g0) {
    char* img = smem + buf * BUFB + imgOff;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int seg = half * 16 + wid * 2 + i;
      const int row = seg * 8 + lane / 8;
      const int kc = (lane % 8) ^ (row & 7);
      auto* gsrc = (const __attribute__((address_space(1))) unsigned int*)
          (const char*)(G + (long)(g0 + row) * ldg + k0 + kc * 8);
      auto* ldst = (__attribute__((address_space(3))) unsigned int*)
          (__attribute__((address_space(3))) char*)(img + seg * 1024);
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);
    }
  };
  auto stage_A = [&](int tile, int half) {
    if (tile < ntiles)
      stage_half(tile & 1, tile * BK, 0, half, A, lda, m0);
  };
  auto stage_B = [&](int tile, int half) {
    if (tile < ntiles)
      stage_half(tile & 1, tile * BK, BM * BKB, half, B, ldb, n0);
  };

  auto lds_frag = [&](int buf, int imgOff, int row, int kk) -> bf16x8 {
    const char* img = smem + buf * BUFB + imgOff;
    return *(const bf16x8*)(img + row * BKB +
                            (((kk * 32 + fkb) * 2) ^ ((row & 7) << 4)));
  };

  // ---- prologue: tiles 0 and 1 fully staged, full drain once -------------
  stage_A(0, 0); stage_A(0, 1); stage_B(0, 0); stage_B(0, 1);
  stage_A(1, 0); stage_A(1, 1); stage_B(1, 0); stage_B(1, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  asm volatile("s_barrier" ::: "memory");

  // ---- main loop: 4 phases per K-tile. The A quadrant for phase q+1 is
  // ds_read right after phase q's MFMA cluster (same buffer, stable until
  // the NEXT tile's stages), so its lgkmcnt wait spans two barriers and the
  // staging section — the per-phase LDS latency is hidden (PMC showed
  // WAIT_INST_LDS ~= busy cycles with reads issued in their own phase).
  bf16x8 afr[2][2][2];                      // [phase parity][mi-pair][kk]
  auto read_quad = [&](int buf_, int q, int pp) {
    #pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int row = wr * 128 + (q * 2 + j) * 16 + fr;
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afr[pp][j][kk] = lds_frag(buf_, 0, row, kk);
    }
  };
  auto read_bfr = [&](int buf_) {
    #pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int row = wc * 64 + ni * 16 + fr;
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfr[ni][kk] = lds_frag(buf_, BM * BKB, row, kk);
    }
  };
  // tile-0 prologue reads (later tiles prefetch during the previous q3)
  read_bfr(0);
  read_quad(0, 0, 0);
  for (int s = 0; s < ntiles; ++s) {
    const int buf = s & 1;
    #pragma unroll
    for (int q = 0; q < 4; ++q) {
      // phase staging (schedule in the header comment)
      if (q == 1) stage_A(s + 1, 0);
      else if (q == 2) stage_A(s + 1, 1);
      else if (q == 3) {
        stage_B(s + 2, 0);
        stage_B(s + 2, 1);
        if (DRAIN)
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      }
      asm volatile("s_barrier" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int j = 0; j < 2; ++j)
        #pragma unroll
        for (int ni = 0; ni < NFRAG; ++ni)
          #pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[q * 2 + j][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[q & 1][j][kk], bfr[ni][kk], acc[q * 2 + j][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (q < 3) {
        read_quad(buf, q + 1, (q + 1) & 1);
      } else if (s + 1 < ntiles) {
        // next tile's frags: content visible after this phase's vmcnt +
        // opening barrier; bfr/afr[0] are dead after q3's MFMA cluster
        read_bfr(buf ^ 1);
        read_quad(buf ^ 1, 0, 0);
      }
      asm volatile("s_barrier" ::: "memory");
    }
  }

  // ---- epilogue: LDS-staged vectorized stores, two row-halves ------------
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // retire leftover glds
  asm volatile("s_barrier" ::: "memory");
  constexpr int CROWB = BN * 2 + 16;        // padded bf16 rows (half: 128 rows)
  char* ct = smem;
  #pragma unroll
  for (int h = 0; h < 2; ++h) {             // wr halves: rows h*128..+128
    if (wr == h) {
      #pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        const int colL = wc * 64 + ni * 16 + (lane & 15);
        const int col = n0 + colL;
        const float bv = bias ? bias[col] : 0.f;
        #pragma unroll
        for (int mi = 0; mi < MFRAG; ++mi) {
          const int rowL = mi * 16 + (lane >> 4) * 4;
          #pragma unroll
          for (int j = 0; j < 4; ++j) {
            float v = acc[mi][ni][j] + bv;
            if (EPI == 1) v = fmaxf(v, 0.f);
            *(bf16*)(ct + (rowL + j) * CROWB + colL * 2) = tobf16(v);
          }
        }
      }
    }
    __syncthreads();
    constexpr int NCH = BN / 8;
    for (int c = tid; c < 128 * NCH; c += 512) {
      const int rowL = c / NCH, ch = c % NCH;
      const long gr = m0 + h * 128 + rowL;
      const int gc = n0 + ch * 8;
      bf16x8 v = *(const bf16x8*)(ct + rowL * CROWB + ch * 16);
      if (addend) {
        const bf16x8 d = *(const bf16x8*)(addend + gr * ldc + gc);
        #pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = tobf16(tofloat(v[e]) + tofloat(d[e]));
      }
      *(bf16x8*)(C + gr * ldc + gc) = v;
    }
    // fused BatchNorm statistics over the staged half (same LDS
    // pre-reduction pattern as gemm.hip's tile epilogue)
    if (stats_sum != nullptr) {
      float* lacc = (float*)(ct + 128 * CROWB + 16);   // [2][BN]
      for (int i = tid; i < 2 * BN; i += 512) lacc[i] = 0.f;
      __syncthreads();
      const int tc = tid & 31, tr = tid >> 5;          // 32 col-chunks x 16
      float s1[8] = {}, s2[8] = {};
      for (int rowL = tr; rowL < 128; rowL += 16) {
        bf16x8 vv = *(const bf16x8*)(ct + rowL * CROWB + tc * 16);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = tofloat(vv[j]);
          s1[j] += f;
          s2[j] += f * f;
        }
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(lacc + tc * 8 + j, s1[j]);
        atomicAdd(lacc + BN + tc * 8 + j, s2[j]);
      }
      __syncthreads();
      for (int i = tid; i < BN; i += 512) {
        atomicAdd(stats_sum + n0 + i, lacc[i]);
        atomicAdd(stats_sumsq + n0 + i, lacc[BN + i]);
      }
    }
    __syncthreads();
  }
}

bool launch_gemm256(const void* A, long lda, const void* B, long ldb, void* C,
                    long ldc, const float* bias, int M, int N, int K, int epi,
                    const void* addend, float* stats_sum, float* stats_sumsq,
                    hipStream_t s) {
  if (M % 256 != 0 || N % 256 != 0 || K % 64 != 0 || K < 256) return false;
  const int grid = (M / 256) * (N / 256);
  constexpr size_t LDS_MAIN = 2 * (256 + 256) * 128;        // 128 KiB
  constexpr size_t LDS_EPI = 128 * (256 * 2 + 16) + 16 + 2 * 256 * 4;
  const size_t lds = LDS_MAIN > LDS_EPI ? LDS_MAIN : LDS_EPI;
  // Grids of >= ~192 of these 128-KiB-LDS workgroups break the counted
  // vmcnt(4) pipeline (r2 measurement: <= 188 blocks 0 failures over
  // hundreds of runs at every K; >= 192 blocks scattered wrong tiles in
  // ~95% of runs, K >= 512, errors across ALL blocks; stream-serialized
  // sub-launches and even hipStreamSynchronize between them do NOT fix it,
  // so it is a residency/scheduling effect, not launch overlap). Big grids
  // take the DRAIN variant: full vmcnt(0) per K-tile — the same schedule
  // with the pipeline depth collapsed, correct by construction.
  const bool drain = grid >= 190;
  #define LO_G256(EPI_, DRAIN_)                                               \
    hipLaunchKernelGGL(HIP_KERNEL_NAME(gemm256_kernel<EPI_, DRAIN_>),         \
                       dim3(grid), dim3(512), lds, s, (const bf16*)A, lda,    \
                       (const bf16*)B, ldb, (bf16*)C, ldc, bias, M, N, K,     \
                       (const bf16*)addend, stats_sum, stats_sumsq)
  if (epi == 1) { if (drain) LO_G256(1, true); else LO_G256(1, false); }
  else          { if (drain) LO_G256(0, true); else LO_G256(0, false); }
  #undef LO_G256
  return true;
}

}  // namespace lo
