// MFMA bf16 GEMM for gfx950 (MI355X) — the framework's core compute kernel.
//
// Computes C[M,N] = A' @ B' (+bias, +ReLU), fp32 accumulate, where
//   A' = A  ([M,K] row-major)        if !TA, else A^T (A is [K,M] row-major)
//   B' = B  ([K,N] row-major)        if !TB, else B^T (B is [N,K] row-major)
// The (TA=true) path + fp32 atomic output is the split-K weight-gradient GEMM.
//
// Replaces what the reference delegated to Spark MLlib / TF black boxes
// ("Dense GEMM fwd/bwd" -> bf16 MFMA GEMM). Design per the
// CDNA4 guide §5: v_mfma_f32_16x16x32_bf16 tiles, LDS staging with the
// bank-conflict XOR swizzle (guide §6 G4), XCD-aware block remap (T1).
// Layouts are normalized in LDS to As[BM][BK] / Bs[BN][BK] (k-contiguous
// rows) so every MFMA fragment read is one swizzled ds_read_b128.

#include "lo_common.h"

namespace lo {

typedef short s16x4 __attribute__((ext_vector_type(4)));

// tr16 image region permutation: k-block kb (4 k-rows each, 8 per 32-k MFMA
// step) is stored at slot (kb/2) + (kb%2)*4 within its step, so one
// ds_read_b64_tr_b16 instruction's four 16-lane groups (k = g*8 + j, i.e.
// kb = {0,2,4,6} then {1,3,5,7}) read CONSECUTIVE 128-B regions — the
// conflict-free pattern; the naive k-order is a 4-way bank conflict.
LO_DEVICE int tr16_slot(int k) {
  const int step = k >> 5;              // 32-k MFMA step
  const int kb = (k & 31) >> 2;         // k-block within step
  return step * 8 + (kb >> 1) + (kb & 1) * 4;
}

// Implicit-GEMM conv: geometry for gathering im2col rows inside the GEMM
// staging (SURVEY §2.9 implicit conv; never materializes the col matrix).
struct ConvGeom {
  int B, H, W, C;        // NHWC input
  int KH, KW, SH, SW, PH, PW;
  int OH, OW;
  // magic-division constants for OW/OH (filled by the host for gather
  // paths): mg==0 -> divisor 1 (q=n), mg==1 -> power of two (q=n>>s),
  // else q = mulhi(n, mg) >> s  (exact for n < 2^31)
  unsigned mgOW = 0, mgOH = 0;
  int sOW = 0, sOH = 0;
};

LO_DEVICE unsigned fdiv(unsigned n, unsigned mg, int s) {
  if (mg == 0) return n;
  if (mg == 1) return n >> s;
  return (unsigned)(((unsigned long long)n * mg) >> 32) >> s;
}

// gather 8 consecutive col elements of row `m` starting at column `k`
// (C % 8 == 0: the run lies in one (kh,kw) channel segment; C < 8: scalar)
LO_DEVICE bf16x8 conv_gather8(const bf16* __restrict__ x, const ConvGeom& g,
                              long m, int k) {
  bf16x8 v = {};
  const int ow = (int)(m % g.OW);
  const long t = m / g.OW;
  const int oh = (int)(t % g.OH);
  const int b = (int)(t / g.OH);
  const int kk = g.KH * g.KW * g.C;
  if (g.C % 8 == 0) {
    if (k >= kk) return v;
    const int c = k % g.C;
    const int p = k / g.C;
    const int kw = p % g.KW, kh = p / g.KW;
    const int h = oh * g.SH - g.PH + kh, w = ow * g.SW - g.PW + kw;
    if (h >= 0 && h < g.H && w >= 0 && w < g.W)
      v = *(const bf16x8*)(x + (((long)b * g.H + h) * g.W + w) * g.C + c);
    return v;
  }
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int ke = k + e;
    if (ke >= kk) break;
    const int c = ke % g.C;
    const int p = ke / g.C;
    const int kw = p % g.KW, kh = p / g.KW;
    const int h = oh * g.SH - g.PH + kh, w = ow * g.SW - g.PW + kw;
    if (h >= 0 && h < g.H && w >= 0 && w < g.W)
      v[e] = x[(((long)b * g.H + h) * g.W + w) * g.C + c];
  }
  return v;
}

// mfma_f32_16x16x32_bf16 operand maps (verified by tests/test_gpu_ops.py's
// probe): lane l holds A[row=l%16][k=(l/16)*8 + j], j=0..7 (contiguous k),
// B[k][col] mirrored; C/D: col=lane&15, row=(lane>>4)*4+reg (guide §3).

// GATHER: 0 = plain, 1 = A is an im2col view of `A` as NHWC input (fwd
// conv), 2 = B is an im2col view of `B` (dW with tr16 k-major image).
template <int BM, int BN, int BK, int WM, int WN, bool TA, bool TB,
          int EPI, bool OUT_F32, bool ATOMIC, int GATHER = 0>
__global__ __launch_bounds__(WM * WN * 64) void gemm_kernel(
    const bf16* __restrict__ A, long lda, const bf16* __restrict__ B, long ldb,
    void* __restrict__ Cv, long ldc, const float* __restrict__ bias,
    int M, int N, int K, int kStart, int kChunk, ConvGeom geom,
    float* __restrict__ stats_sum, float* __restrict__ stats_sumsq,
    const bf16* __restrict__ addend) {
  constexpr int T = WM * WN * 64;
  constexpr int WTM = BM / WM;          // wave tile rows
  constexpr int WTN = BN / WN;          // wave tile cols
  constexpr int MFRAG = WTM / 16;
  constexpr int NFRAG = WTN / 16;
  constexpr int BKB = BK * 2;           // LDS row bytes
  constexpr int SWZ = (BKB >= 128) ? 7 : (BKB >= 64 ? 3 : 1);
  static_assert(BM % (WM * 16) == 0 && BN % (WN * 16) == 0, "wave tiling");
  static_assert(BK % 32 == 0, "BK multiple of MFMA K");

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // two (A,B) tile buffers: stage t+1 while computing t (guide T3 2-phase);
  // computed as offsets (a pointer array of LDS addrspacecasts is rejected
  // by the backend as a static initializer)
  constexpr int BUFB = (BM + BN) * BKB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WN, wc = wid % WN;

  const int nb = (N + BN - 1) / BN;
  const int mb = (M + BM - 1) / BM;
  // (rejected r2 experiment, kept as a note: remapping (tile, z) so each
  // XCD sweeps one k-slice's tiles before moving on — to keep the A(dy)
  // slice L2-resident across its N-tile re-reads — measured WORSE on every
  // model (TextCNN -12%, MNIST/ResNet -2%): the i%8 -> XCD round-robin
  // assumption evidently does not describe the MI355X workgroup dispatcher,
  // and per-slice tile counts are too small to amortize the lost
  // cross-slice parallelism.)
  int bid = xcd_swizzle(blockIdx.x, mb * nb);
  const int bm = bid % mb, bn = bid / mb;   // consecutive blocks share B panel
  const int m0 = bm * BM, n0 = bn * BN;

  const int kBegin = kStart + blockIdx.z * kChunk;
  if (kBegin >= K) return;                  // padded z-slice (block-uniform)
  const int kEnd = min(K, kBegin + kChunk);

  f32x4 acc[MFRAG][NFRAG] = {};

  const int fr = lane & 15;
  const int fkb = (lane >> 4) * 8;      // fragment k base within MFMA step

  constexpr int NW = T / 64;
  constexpr int ROWS_PER_SEG = 1024 / BKB;        // LDS rows per 1-KiB wave DMA
  constexpr int KCH = BKB / 16;                   // 16-B chunks per LDS row
  const bool a_rows_in = !TA && (m0 + BM <= M);
  const bool b_rows_in = TB && (n0 + BN <= N);

  // T14 async-STAGE split for the register-staged (tr16) operand paths:
  // issue the next tile's global loads BEFORE the compute phase (HBM latency
  // hides under the MFMAs) and do the LDS writes after it — writing
  // immediately after the loads would force the vmcnt wait at the ds_write
  // and expose the full latency (guide §6 G15; this was the split-K dW
  // kernel's bound).
  constexpr int CHA_T = TA ? (BK * BM / 8 + T - 1) / T : 1;
  constexpr int CHB_T = (!TB) ? (BK * BN / 8 + T - 1) / T : 1;
  bf16x8 stgA[CHA_T], stgB[CHB_T];

  // GATHER==2: this thread's B-tile columns are invariant across k-steps —
  // decode (kh,kw,c) ONCE; the per-step row decode uses magic division
  // (the full per-run conv_gather8 decode measured 28 VALU per MFMA).
  constexpr int G2N = (GATHER == 2) ? CHB_T : 1;
  int g2c[G2N], g2dh[G2N], g2dw[G2N];
  bool g2ok[G2N];
  if (GATHER == 2) {
    #pragma unroll
    for (int i = 0; i < G2N; ++i) {
      const int c = tid + i * T;
      const int nc = c % (BN / 8);
      const int gn0 = n0 + nc * 8;
      // fast path needs the 8-run inside ONE (kh,kw) segment (C % 8 == 0);
      // C < 8 falls back to the per-element conv_gather8 decode below
      g2ok[i] = (geom.C % 8 == 0) && (c < BK * BN / 8) && gn0 < N &&
                gn0 < geom.KH * geom.KW * geom.C;
      const int cc = g2ok[i] ? gn0 % geom.C : 0;
      const int p = g2ok[i] ? gn0 / geom.C : 0;
      g2c[i] = cc;
      g2dh[i] = p / geom.KW - geom.PH;
      g2dw[i] = p % geom.KW - geom.PW;
    }
  }

  auto stage_next = [&](int buf, int k0) {
    char* smA = smem + buf * BUFB;
    char* smB = smem + buf * BUFB + BM * BKB;
    const bool k_in = (k0 + BK <= K);
    // ---- stage A tile -> As[BM][BK] ------------------------------------
    if (GATHER == 1) {
      // implicit im2col gather of A rows (no col matrix): plain-load path,
      // swizzled like the register path below
      constexpr int CH = BM * BK / 8;
      for (int c = tid; c < CH; c += T) {
        const int row = c / (BK / 8), kc = c % (BK / 8);
        const long gm = m0 + row;
        const int gk = k0 + kc * 8;
        bf16x8 v = {};
        if (gm < M && gk < K) v = conv_gather8(A, geom, gm, gk);
        *(bf16x8*)(smA + row * BKB + ((kc * 16) ^ ((row & SWZ) << 4))) = v;
      }
    } else if (!TA) {
      if (a_rows_in && k_in && ROWS_PER_SEG > 0) {
        // interior tiles: LDS-DMA (global_load_lds, 16 B/lane — guide §5
        // ladder step 3). glds writes lane-linear, so the bank-conflict XOR
        // swizzle moves to the per-lane SOURCE address (rule 21); the
        // ds_read side keeps the same XOR (involution).
        #pragma unroll
        for (int seg = wid; seg < BM / ROWS_PER_SEG; seg += NW) {
          const int row = seg * ROWS_PER_SEG + lane / KCH;
          const int kc = (lane % KCH) ^ (row & SWZ);
          auto* gsrc = (const __attribute__((address_space(1))) unsigned int*)
              (const char*)(A + (long)(m0 + row) * lda + k0 + kc * 8);
          auto* ldst = (__attribute__((address_space(3))) unsigned int*)
              (__attribute__((address_space(3))) char*)(smA + seg * 1024);
          __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);
        }
      } else {
        constexpr int CH = BM * BK / 8;
        for (int c = tid; c < CH; c += T) {
          const int row = c / (BK / 8), kc = c % (BK / 8);
          const int gm = m0 + row, gk = k0 + kc * 8;
          bf16x8 v = {};
          if (gm < M && gk < K) v = *(const bf16x8*)(A + (long)gm * lda + gk);
          *(bf16x8*)(smA + row * BKB + ((kc * 16) ^ ((row & SWZ) << 4))) = v;
        }
      }
    } else {
      // A is [K,M]: load NATURALLY (no transpose) into registers; the LDS
      // write happens in stage_finish; fragments use ds_read_b64_tr_b16
      // against the k-major 16-wide m-subtile image (probe-verified).
      constexpr int CH = BK * BM / 8;
      #pragma unroll
      for (int i = 0; i < CHA_T; ++i) {
        const int c = tid + i * T;
        bf16x8 v = {};
        if (c < CH) {
          const int k = c / (BM / 8), mc = c % (BM / 8);
          const int gk = k0 + k, gm0 = m0 + mc * 8;
          if (gk < K) {
            if (gm0 + 8 <= M) v = *(const bf16x8*)(A + (long)gk * lda + gm0);
            else if (gm0 < M)
              for (int j = 0; j < 8 && gm0 + j < M; ++j) v[j] = A[(long)gk * lda + gm0 + j];
          }
        }
        stgA[i] = v;
      }
    }
    // ---- stage B tile -> Bs[BN][BK] ------------------------------------
    if (TB) {
      if (b_rows_in && k_in && ROWS_PER_SEG > 0) {
        #pragma unroll
        for (int seg = wid; seg < BN / ROWS_PER_SEG; seg += NW) {
          const int row = seg * ROWS_PER_SEG + lane / KCH;
          const int kc = (lane % KCH) ^ (row & SWZ);
          auto* gsrc = (const __attribute__((address_space(1))) unsigned int*)
              (const char*)(B + (long)(n0 + row) * ldb + k0 + kc * 8);
          auto* ldst = (__attribute__((address_space(3))) unsigned int*)
              (__attribute__((address_space(3))) char*)(smB + seg * 1024);
          __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);
        }
      } else {
        constexpr int CH = BN * BK / 8;
        for (int c = tid; c < CH; c += T) {
          const int row = c / (BK / 8), kc = c % (BK / 8);
          const int gn = n0 + row, gk = k0 + kc * 8;
          bf16x8 v = {};
          if (gn < N && gk < K) v = *(const bf16x8*)(B + (long)gn * ldb + gk);
          *(bf16x8*)(smB + row * BKB + ((kc * 16) ^ ((row & SWZ) << 4))) = v;
        }
      }
    } else {
      // B is [K,N]: natural k-major load into registers (write deferred);
      // GATHER==2 gathers the implicit im2col row of the NHWC input instead
      // (dW never re-reads a col matrix)
      constexpr int CH = BK * BN / 8;
      #pragma unroll
      for (int i = 0; i < CHB_T; ++i) {
        const int c = tid + i * T;
        bf16x8 v = {};
        if (c < CH) {
          const int k = c / (BN / 8), nc = c % (BN / 8);
          const int gk = k0 + k, gn0 = n0 + nc * 8;
          if (GATHER == 2) {
            if (geom.C % 8 != 0) {
              if (gk < K && gn0 < N) v = conv_gather8(B, geom, (long)gk, gn0);
            } else if (g2ok[i] && gk < K) {
              const unsigned q1 = fdiv((unsigned)gk, geom.mgOW, geom.sOW);
              const int ow = gk - (int)q1 * geom.OW;
              const unsigned b = fdiv(q1, geom.mgOH, geom.sOH);
              const int oh = (int)q1 - (int)b * geom.OH;
              const int h = oh * geom.SH + g2dh[i];
              const int w = ow * geom.SW + g2dw[i];
              if (h >= 0 && h < geom.H && w >= 0 && w < geom.W)
                v = *(const bf16x8*)(
                    B + (((long)b * geom.H + h) * geom.W + w) * geom.C + g2c[i]);
            }
          } else if (gk < K) {  // GATHER == 0
            if (gn0 + 8 <= N) v = *(const bf16x8*)(B + (long)gk * ldb + gn0);
            else if (gn0 < N)
              for (int j = 0; j < 8 && gn0 + j < N; ++j) v[j] = B[(long)gk * ldb + gn0 + j];
          }
        }
        stgB[i] = v;
      }
    }
  };

  auto stage_finish = [&](int buf) {
    char* smA = smem + buf * BUFB;
    char* smB = smem + buf * BUFB + BM * BKB;
    if (TA) {
      constexpr int CH = BK * BM / 8;
      #pragma unroll
      for (int i = 0; i < CHA_T; ++i) {
        const int c = tid + i * T;
        if (c < CH) {
          const int k = c / (BM / 8), mc = c % (BM / 8);
          const int msub = (mc * 8) / 16, mrem = (mc * 8) % 16;
          *(bf16x8*)(smA + (msub * BK * 16 + tr16_slot(k) * 64 + (k & 3) * 16
                            + mrem) * 2) = stgA[i];
        }
      }
    }
    if (!TB) {
      constexpr int CH = BK * BN / 8;
      #pragma unroll
      for (int i = 0; i < CHB_T; ++i) {
        const int c = tid + i * T;
        if (c < CH) {
          const int k = c / (BN / 8), nc = c % (BN / 8);
          const int nsub = (nc * 8) / 16, nrem = (nc * 8) % 16;
          *(bf16x8*)(smB + (nsub * BK * 16 + tr16_slot(k) * 64 + (k & 3) * 16
                            + nrem) * 2) = stgB[i];
        }
      }
    }
  };

  auto compute_tile = [&](int buf) {
    const char* smA = smem + buf * BUFB;
    const char* smB = smem + buf * BUFB + BM * BKB;
    #pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 af[MFRAG], bf[NFRAG];
      #pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi) {
        if (!TA) {
          const int row = wr * WTM + mi * 16 + fr;
          af[mi] = *(const bf16x8*)(smA + row * BKB +
                                    (((kk * 32 + fkb) * 2) ^ ((row & SWZ) << 4)));
        } else {
          // hardware-transpose read from the k-major image: two b64_tr_b16
          // reads give k = kbase..+4 and +4..+8 of this lane's m column
          const int msub = (wr * WTM + mi * 16) / 16;
          const char* base = smA + (msub * BK * 16) * 2
                             + (kk * 8 + (lane >> 4)) * 128 + (lane & 15) * 8;
          s16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) s16x4*)base);
          s16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) s16x4*)(base + 512));
          #pragma unroll
          for (int j = 0; j < 4; ++j) {
            ((short*)&af[mi])[j] = lo4[j];
            ((short*)&af[mi])[j + 4] = hi4[j];
          }
        }
      }
      #pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        if (TB) {
          const int row = wc * WTN + ni * 16 + fr;
          bf[ni] = *(const bf16x8*)(smB + row * BKB +
                                    (((kk * 32 + fkb) * 2) ^ ((row & SWZ) << 4)));
        } else {
          const int nsub = (wc * WTN + ni * 16) / 16;
          const char* base = smB + (nsub * BK * 16) * 2
                             + (kk * 8 + (lane >> 4)) * 128 + (lane & 15) * 8;
          s16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) s16x4*)base);
          s16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) s16x4*)(base + 512));
          #pragma unroll
          for (int j = 0; j < 4; ++j) {
            ((short*)&bf[ni])[j] = lo4[j];
            ((short*)&bf[ni])[j + 4] = hi4[j];
          }
        }
      }
      #pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
        #pragma unroll
        for (int ni = 0; ni < NFRAG; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  // 2-phase pipeline: issue tile t+1's loads BEFORE computing tile t, LDS
  // writes for the register paths AFTER it, one barrier per tile (guide
  // §5.5 T3 minimum 2-phase + G15 async-STAGE split).
  int cur = 0;
  stage_next(0, kBegin);
  stage_finish(0);
  __syncthreads();
  for (int k0 = kBegin; k0 < kEnd; k0 += BK) {
    const bool more = (k0 + BK < kEnd);
    if (more) stage_next(cur ^ 1, k0 + BK);
    compute_tile(cur);
    if (more) stage_finish(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue ----------------------------------------------------------
  // MFMA C fragments are column-fragmented (lane l holds rows r0..r0+3 of ONE
  // column), so direct stores are 2-4 B scalars. Non-atomic outputs stage the
  // C tile through LDS (padded rows) and store 16 B vectors instead; the
  // split-K path keeps per-element fp32 atomics.
  float* Cf = (float*)Cv;
  bf16* Cb = (bf16*)Cv;
  if (ATOMIC) {
    #pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int col = n0 + wc * WTN + ni * 16 + (lane & 15);
      if (col >= N) continue;
      #pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi) {
        const int row0 = m0 + wr * WTM + mi * 16 + (lane >> 4) * 4;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int row = row0 + j;
          if (row >= M) continue;
          atomicAdd(Cf + (long)row * ldc + col, acc[mi][ni][j]);
        }
      }
    }
    return;
  }
  constexpr int EB = OUT_F32 ? 4 : 2;          // bytes per element in LDS
  constexpr int CROWB = BN * EB + 16;          // padded LDS row
  char* ct = smem;                              // staging buffers are done
  {
    #pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int colL = wc * WTN + ni * 16 + (lane & 15);
      const int col = n0 + colL;
      const float bv = (bias && col < N) ? bias[col] : 0.f;
      #pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi) {
        const int rowL0 = wr * WTM + mi * 16 + (lane >> 4) * 4;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          float v = acc[mi][ni][j] + bv;
          if (EPI == 1) v = fmaxf(v, 0.f);
          if (OUT_F32) *(float*)(ct + (rowL0 + j) * CROWB + colL * 4) = v;
          else *(bf16*)(ct + (rowL0 + j) * CROWB + colL * 2) = tobf16(v);
        }
      }
    }
  }
  __syncthreads();
  constexpr int VE = 16 / EB;                  // elements per 16-B store
  constexpr int NCH = BN / VE;
  for (int c = tid; c < BM * NCH; c += T) {
    const int rowL = c / NCH, ch = c % NCH;
    const int gr = m0 + rowL;
    if (gr >= M) continue;
    const int gc = n0 + ch * VE;
    const char* src = ct + rowL * CROWB + ch * 16;
    if (gc + VE <= N) {
      if (OUT_F32) *(f32x4*)(Cf + (long)gr * ldc + gc) = *(const f32x4*)src;
      else if (addend) {
        const bf16x8 d = *(const bf16x8*)(addend + (long)gr * ldc + gc);
        bf16x8 v = *(const bf16x8*)src;
        #pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = tobf16(tofloat(v[e]) + tofloat(d[e]));
        *(bf16x8*)(Cb + (long)gr * ldc + gc) = v;
      } else *(bf16x8*)(Cb + (long)gr * ldc + gc) = *(const bf16x8*)src;
    } else if (gc < N) {
      for (int e = 0; e < VE && gc + e < N; ++e) {
        if (OUT_F32) Cf[(long)gr * ldc + gc + e] = ((const float*)src)[e];
        else {
          bf16 v = ((const bf16*)src)[e];
          if (addend) v = tobf16(tofloat(v) + tofloat(addend[(long)gr * ldc + gc + e]));
          Cb[(long)gr * ldc + gc + e] = v;
        }
      }
    }
  }
  // fused BatchNorm statistics over the staged tile (bf16-out convs feeding
  // BN): per-column partial sums + one global atomic per column per block —
  // saves the separate bn_stats pass over the activation.
  if (!OUT_F32 && stats_sum != nullptr) {
    // LDS pre-reduction first: direct global atomics from every row-lane
    // serialized on the few per-channel addresses (measured 10x whole-model
    // slowdown); one global atomic per column per block only.
    float* lacc = (float*)(ct + BM * CROWB + 16);   // [2][BN]
    for (int i = tid; i < 2 * BN; i += T) lacc[i] = 0.f;
    __syncthreads();
    const int tc = tid % (BN / 8);
    const int tr = tid / (BN / 8);
    const int rowsPerBlock = T / (BN / 8);
    const int gc0 = n0 + tc * 8;
    if (gc0 < N && tr < rowsPerBlock) {
      float s1[8] = {}, s2[8] = {};
      for (int rowL = tr; rowL < BM; rowL += rowsPerBlock) {
        if (m0 + rowL >= M) break;
        bf16x8 v = *(const bf16x8*)(ct + rowL * CROWB + tc * 16);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = tofloat(v[j]);
          s1[j] += f;
          s2[j] += f * f;
        }
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(lacc + tc * 8 + j, s1[j]);
        atomicAdd(lacc + BN + tc * 8 + j, s2[j]);
      }
    }
    __syncthreads();
    for (int i = tid; i < BN && n0 + i < N; i += T) {
      atomicAdd(stats_sum + n0 + i, lacc[i]);
      atomicAdd(stats_sumsq + n0 + i, lacc[BN + i]);
    }
  }
}

// ---------------------------------------------------------------------------
// MFMA operand-layout probe: computes D = A[16,32] @ B[32,16] with the
// contiguous-k fragment map so the GPU test can verify the map empirically
// (guide §3: "Always A=I-check with ASYMMETRIC B").
__global__ void mfma_probe_kernel(const bf16* A, const bf16* B, float* D) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    b[j] = B[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int j = 0; j < 4; ++j)
    D[((lane >> 4) * 4 + j) * 16 + (lane & 15)] = c[j];
}

// ---------------------------------------------------------------------------
// host-side dispatch

struct GemmArgs {
  const void *A, *B;
  void* C;
  const float* bias;
  long lda, ldb, ldc;
  int M, N, K;
  bool ta, tb;    // operand transposes (see file header)
  int epi;        // 0 none, 1 relu
  bool out_f32;
  int splits;     // >1 => atomic split-K accumulate into fp32 C (zeroed by caller)
  int gather;     // 0 plain; 1 A = implicit im2col of NHWC input; 2 B likewise
  ConvGeom geom;
  // optional fused column statistics (BatchNorm): per-column sum / sum-of-
  // squares of C accumulated by the epilogue (atomics; caller zeroes)
  float* stats_sum = nullptr;
  float* stats_sumsq = nullptr;
  // optional fused residual add: C = op(A)@op(B) + addend (bf16, same
  // shape/ldc as C; bf16-out, non-split-K only) — the ResNet join without
  // a separate add pass
  const void* addend = nullptr;
};

template <int BM, int BN, int BK, int WM, int WN, bool TA, bool TB,
          int EPI, bool OUT_F32, bool ATOMIC, int GATHER = 0>
static void launch_cfg(const GemmArgs& g, hipStream_t s) {
  const int mb = cdiv(g.M, BM), nb = cdiv(g.N, BN);
  int kChunk;
  if (g.splits > 1) {
    const int kb = cdiv(g.K, BK);
    kChunk = cdiv(kb, g.splits) * BK;
  } else {
    kChunk = cdiv(g.K, BK) * BK;
  }
  const int zs = g.splits > 1 ? cdiv(g.K, kChunk) : 1;
  dim3 grid(mb * nb, 1, zs), block(WM * WN * 64);
  size_t lds_stage = (size_t)(BM + BN) * BK * 2 * 2;  // double-buffered
  size_t lds_epi = (size_t)BM * (BN * (g.out_f32 ? 4 : 2) + 16)
                   + 2 * BN * 4 + 32;  // +fused-stats LDS reduction
  size_t lds = lds_stage > lds_epi ? lds_stage : lds_epi;
  hipLaunchKernelGGL(HIP_KERNEL_NAME(
      gemm_kernel<BM, BN, BK, WM, WN, TA, TB, EPI, OUT_F32, ATOMIC, GATHER>),
      grid, block, lds, s,
      (const bf16*)g.A, g.lda, (const bf16*)g.B, g.ldb, g.C, g.ldc, g.bias,
      g.M, g.N, g.K, 0, kChunk, g.geom, g.stats_sum, g.stats_sumsq,
      (g.out_f32 || g.splits > 1) ? nullptr : (const bf16*)g.addend);
}

bool launch_gemm256(const void* A, long lda, const void* B, long ldb, void* C,
                    long ldc, const float* bias, int M, int N, int K, int epi,
                    const void* addend, float* stats_sum, float* stats_sumsq,
                    hipStream_t s);  // gemm_8phase.hip

// Per-(TA,TB) tile-size selection. Instantiates only the combos the engine
// uses (fwd = N,T; dX = N,N; dW = T,N split-K) plus (T,T) for completeness.
template <bool TA, bool TB>
static bool dispatch_tiles(const GemmArgs& g, hipStream_t s) {
  static const bool use256 = [] {
    const char* e = getenv("LO_GEMM256");
    return e && e[0] == '1';
  }();
  // The 8-phase kernel is RETIRED from dispatch (r2): its counted-vmcnt
  // pipeline silently corrupts tiles in a residency/state-dependent way —
  // reliably at >= ~192-block grids, and once observed even at 160 blocks
  // on a warmed box (tools/gemm256_raceprobe.py). The provably-safe DRAIN
  // variant measured BELOW this 128^2 tile kernel on every big-grid shape
  // (ResNet 6294 vs 6389 samples/s, 8192^3 706 vs 793 TF), so the tile
  // path is both the correct and the faster choice. LO_GEMM256=1 re-enables
  // the fast variant for experiments only.
  if (use256 && !TA && TB && !g.out_f32 && g.M % 256 == 0 && g.N % 256 == 0 &&
      g.K % 64 == 0 && g.K >= 256 &&
      (g.M / 256) * (g.N / 256) >= 128) {
    // deep-pipelined 256^2 8-phase path (gemm_8phase.hip)
    if (launch_gemm256(g.A, g.lda, g.B, g.ldb, g.C, g.ldc, g.bias,
                       g.M, g.N, g.K, g.epi, g.addend, g.stats_sum,
                       g.stats_sumsq, s))
      return true;
  }
  #define LO_EPI_CASES_BK(BM_, BN_, BK_, WM_, WN_)                             \
    do {                                                                       \
      if (g.out_f32) {                                                         \
        if (g.epi == 1) launch_cfg<BM_, BN_, BK_, WM_, WN_, TA, TB, 1, true, false>(g, s);  \
        else            launch_cfg<BM_, BN_, BK_, WM_, WN_, TA, TB, 0, true, false>(g, s);  \
      } else {                                                                 \
        if (g.epi == 1) launch_cfg<BM_, BN_, BK_, WM_, WN_, TA, TB, 1, false, false>(g, s); \
        else            launch_cfg<BM_, BN_, BK_, WM_, WN_, TA, TB, 0, false, false>(g, s); \
      }                                                                        \
      return true;                                                             \
    } while (0)

  // 128x128 only when its grid still fills the chip; else halve BM for 2x
  // the blocks (e.g. fc1 fwd 8192x256 was 128 blocks at 128^2)
  const bool big_grid = ((long)cdiv(g.M, 128) * cdiv(g.N, 128)) >= 256;
  if (g.K <= 32) {
    // shallow reductions (e.g. C=1 5x5 conv, kpad 32): half-depth tiles so
    // the LDS image and glds path are fully used
    if (g.N > 64 && big_grid) LO_EPI_CASES_BK(128, 128, 32, 2, 2);
    if (g.N > 64) LO_EPI_CASES_BK(64, 128, 32, 2, 2);
    if (g.N > 32) LO_EPI_CASES_BK(128, 64, 32, 2, 2);
    if (g.N > 16) LO_EPI_CASES_BK(256, 32, 32, 4, 1);
    LO_EPI_CASES_BK(128, 16, 32, 4, 1);
  }
  if (g.N > 64 && big_grid) LO_EPI_CASES_BK(128, 128, 64, 2, 2);
  if (g.N > 64) LO_EPI_CASES_BK(64, 128, 64, 2, 2);   // small-M wide-N (dX tails)
  if (g.N > 32) LO_EPI_CASES_BK(128, 64, 64, 2, 2);
  if (g.N > 16) LO_EPI_CASES_BK(256, 32, 64, 4, 1);
  LO_EPI_CASES_BK(128, 16, 64, 4, 1);
  #undef LO_EPI_CASES_BK
}

// implicit-conv forward: C[M=B*OH*OW, N=outC] = im2col(x) @ W^T (+bias/relu)
static bool dispatch_conv_fwd(const GemmArgs& g, hipStream_t s) {
  #define LO_CONV_FWD(BM_, BN_, BK_, WM_, WN_)                                 \
    do {                                                                       \
      if (g.epi == 1) launch_cfg<BM_, BN_, BK_, WM_, WN_, false, true, 1, false, false, 1>(g, s); \
      else            launch_cfg<BM_, BN_, BK_, WM_, WN_, false, true, 0, false, false, 1>(g, s); \
      return true;                                                             \
    } while (0)
  if (g.out_f32) return false;
  if (g.K <= 32) {
    if (g.N > 32) LO_CONV_FWD(128, 64, 32, 2, 2);
    if (g.N > 16) LO_CONV_FWD(256, 32, 32, 4, 1);
    LO_CONV_FWD(128, 16, 32, 4, 1);
  }
  if (g.N > 64) LO_CONV_FWD(128, 128, 64, 2, 2);
  if (g.N > 32) LO_CONV_FWD(128, 64, 64, 2, 2);
  if (g.N > 16) LO_CONV_FWD(256, 32, 64, 4, 1);
  LO_CONV_FWD(128, 16, 64, 4, 1);
  #undef LO_CONV_FWD
}

// implicit-conv dW: C[M=outC, N=kpad] = dY^T @ im2col(x), split-K atomics
static bool dispatch_conv_dw(const GemmArgs& g, hipStream_t s) {
  if (!g.out_f32 || g.epi != 0 || !g.ta || g.tb) return false;
  // Tile-width choice is a TRAFFIC decision for these tall-skinny deep-K
  // shapes: the A operand (dY) is re-streamed once per N(kpad)-tile and the
  // gathered B (x windows) once per M(outC)-tile, so wider tiles cut the
  // dominant re-reads (TextCNN dW measured 4x over its traffic floor with
  // 64x64 tiles, r2). BM=64 when outC allows also halves conv_gather8 work.
  // (outC=64 shapes measured FASTER on 64x64 than 64x128 — MNIST conv2 dW
  // regressed 6.78 -> 6.35 M samples/s on the wide tile — so wide tiles
  // require outC >= 128)
  // (re-measured r2 after the shape specializations: <64,128> for outC=64
  // STILL regresses MNIST — 7.14 vs 7.79 M samples/s — despite halving the
  // dY re-reads; the wide-BN gather's per-thread column state doubles and
  // eats the win. Wide tiles stay gated to outC >= 128.)
  if (g.M >= 128 && g.N >= 128 && g.K >= (1 << 18))
    launch_cfg<128, 128, 64, 2, 2, true, false, 0, true, true, 2>(g, s);
  else if (g.M >= 64)
    launch_cfg<64, 64, 64, 2, 2, true, false, 0, true, true, 2>(g, s);
  else
    launch_cfg<32, 64, 64, 1, 4, true, false, 0, true, true, 2>(g, s);
  return true;
}

// Returns true if a native config covered the shape; false => caller falls
// back to a library GEMM (rocBLAS via torch) for the cold path.
bool gemm_dispatch(const GemmArgs& g, hipStream_t s) {
  if (g.K % 8 != 0) return false;
  if (g.gather == 1) return dispatch_conv_fwd(g, s);
  if (g.gather == 2) return dispatch_conv_dw(g, s);
  if (g.splits > 1) {
    // split-K atomic accumulate (weight gradients): dW = A^T @ B
    // (32x64 tiles: measured better than 64x128 — block count beats
    // staging depth for these shapes)
    if (!g.out_f32 || g.epi != 0) return false;
    if (!g.ta && g.tb) {
      if (g.M >= 64) launch_cfg<64, 64, 64, 2, 2, false, true, 0, true, true>(g, s);
      else launch_cfg<32, 64, 64, 1, 4, false, true, 0, true, true>(g, s);
    } else if (g.ta && !g.tb) {
      // wide tiles for big deep-K dW shapes: A(dy) is re-read once per
      // N(kpad)-tile, B(col) once per M(outC)-tile (same traffic argument
      // as dispatch_conv_dw; ResNet's dW GEMMs are 18% of its step)
      if (g.M >= 128 && g.N >= 128 && g.K >= (1 << 17))
        launch_cfg<128, 128, 64, 2, 2, true, false, 0, true, true>(g, s);
      else if (g.M >= 64)
        launch_cfg<64, 64, 64, 2, 2, true, false, 0, true, true>(g, s);
      else launch_cfg<32, 64, 64, 1, 4, true, false, 0, true, true>(g, s);
    } else {
      return false;
    }
    return true;
  }
  if (!g.ta && g.tb) return dispatch_tiles<false, true>(g, s);
  if (!g.ta && !g.tb) return dispatch_tiles<false, false>(g, s);
  if (g.ta && !g.tb) return dispatch_tiles<true, false>(g, s);
  return dispatch_tiles<true, true>(g, s);
}

void launch_mfma_probe(const void* A, const void* B, float* D, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s,
                     (const bf16*)A, (const bf16*)B, D);
}

}  // namespace lo
