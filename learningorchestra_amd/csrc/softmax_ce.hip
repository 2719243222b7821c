// Fused softmax + cross-entropy, forward AND backward in one kernel pass —
// SURVEY.md §2.9 "Softmax + cross-entropy -> fused softmax-CE fwd/bwd".
//
// logits [M, C] bf16 (C <= 32: one thread per row, register-resident;
// larger C: one wave per row with shuffle reductions). Cvalid <= C supports
// the padded-class trick (the engine pads the classifier head to a multiple
// of 8 for aligned GEMMs; padded logits are treated as -inf and get 0 grad).
//
// Outputs in a single pass:
//   dlogits[m,c] = (softmax(m,c) - onehot) * gscale     (bf16)
//   loss_sum    += -log softmax(m, label_m)             (fp32 atomic)
//   correct     += (argmax == label)                    (int atomic)

#include "lo_common.h"

namespace lo {

__global__ void softmax_ce_small_kernel(const bf16* __restrict__ logits,
                                        const long* __restrict__ labels,
                                        bf16* __restrict__ dlogits,
                                        float* __restrict__ loss_sum,
                                        int* __restrict__ correct,
                                        long M, int C, int Cvalid, float gscale) {
  float wave_loss = 0.f;
  int wave_correct = 0;
  for (long r = (long)blockIdx.x * blockDim.x + threadIdx.x; r < M;
       r += (long)gridDim.x * blockDim.x) {
    float v[32];
    float mx = -3.0e38f;
    for (int c = 0; c < Cvalid; ++c) {
      v[c] = tofloat(logits[r * C + c]);
      mx = fmaxf(mx, v[c]);
    }
    float sum = 0.f;
    int am = 0;
    float amv = -3.0e38f;
    for (int c = 0; c < Cvalid; ++c) {
      if (v[c] > amv) { amv = v[c]; am = c; }
      v[c] = __expf(v[c] - mx);
      sum += v[c];
    }
    const float inv = 1.f / sum;
    const int y = (int)labels[r];
    for (int c = 0; c < Cvalid; ++c) {
      const float p = v[c] * inv;
      dlogits[r * C + c] = tobf16((p - (c == y ? 1.f : 0.f)) * gscale);
    }
    for (int c = Cvalid; c < C; ++c) dlogits[r * C + c] = bf16(0.f);
    wave_loss += -(__logf(v[y] * inv));
    wave_correct += (am == y);
  }
  // one atomic per wave (guide G12)
  for (int off = 32; off > 0; off >>= 1) {
    wave_loss += __shfl_down(wave_loss, off);
    wave_correct += __shfl_down(wave_correct, off);
  }
  if ((threadIdx.x & 63) == 0) {
    if (loss_sum) atomicAdd(loss_sum, wave_loss);
    if (correct) atomicAdd(correct, wave_correct);
  }
}

// wave-per-row path for C in (32, 8192]; C % 8 == 0 required (engine pads)
__global__ void softmax_ce_wave_kernel(const bf16* __restrict__ logits,
                                       const long* __restrict__ labels,
                                       bf16* __restrict__ dlogits,
                                       float* __restrict__ loss_sum,
                                       int* __restrict__ correct,
                                       long M, int C, int Cvalid, float gscale) {
  const int lane = threadIdx.x & 63;
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int nwaves = (gridDim.x * blockDim.x) >> 6;
  for (long r = wid; r < M; r += nwaves) {
    float mx = -3.0e38f;
    int am = 0;
    for (int c = lane; c < Cvalid; c += 64) {
      const float f = tofloat(logits[r * C + c]);
      if (f > mx) { mx = f; am = c; }
    }
    // max + argmax reduce across lanes
    for (int off = 32; off > 0; off >>= 1) {
      const float omx = __shfl_down(mx, off);
      const int oam = __shfl_down(am, off);
      if (omx > mx) { mx = omx; am = oam; }
    }
    mx = __shfl(mx, 0);
    am = __shfl(am, 0);
    float sum = 0.f;
    for (int c = lane; c < Cvalid; c += 64)
      sum += __expf(tofloat(logits[r * C + c]) - mx);
    for (int off = 32; off > 0; off >>= 1) sum += __shfl_down(sum, off);
    sum = __shfl(sum, 0);
    const float inv = 1.f / sum;
    const int y = (int)labels[r];
    for (int c = lane; c < C; c += 64) {
      float g = 0.f;
      if (c < Cvalid) {
        const float p = __expf(tofloat(logits[r * C + c]) - mx) * inv;
        g = (p - (c == y ? 1.f : 0.f)) * gscale;
      }
      dlogits[r * C + c] = tobf16(g);
    }
    if (lane == 0) {
      if (loss_sum)
        atomicAdd(loss_sum, -(tofloat(logits[r * C + y]) - mx - __logf(sum)));
      if (correct) atomicAdd(correct, (int)(am == y));
    }
  }
}

void launch_softmax_ce(const void* logits, const void* labels, void* dlogits,
                       void* loss_sum, void* correct, long M, int C, int Cvalid,
                       float gscale, hipStream_t s) {
  const int block = 256;
  if (Cvalid <= 32) {
    const int grid = (int)min((M + block - 1) / block, (long)2048);
    hipLaunchKernelGGL(softmax_ce_small_kernel, dim3(grid), dim3(block), 0, s,
                       (const bf16*)logits, (const long*)labels, (bf16*)dlogits,
                       (float*)loss_sum, (int*)correct, M, C, Cvalid, gscale);
  } else {
    const long waves = M;
    const int grid = (int)min((waves * 64 + block - 1) / block, (long)2048);
    hipLaunchKernelGGL(softmax_ce_wave_kernel, dim3(grid), dim3(block), 0, s,
                       (const bf16*)logits, (const long*)labels, (bf16*)dlogits,
                       (float*)loss_sum, (int*)correct, M, C, Cvalid, gscale);
  }
}

}  // namespace lo
