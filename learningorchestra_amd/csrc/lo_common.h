// Common device helpers for the learningorchestra_amd CDNA4 (gfx950) kernels.
// Hand-written HIP for MI355X: wave64, MFMA matrix cores, LDS XOR-swizzles.
// No CUDA-compat shims, no hipify output — gfx950-only by design.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define LO_DEVICE __device__ __forceinline__

namespace lo {

typedef __bf16 bf16;
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));   // 16 B / lane
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

constexpr int WAVE = 64;  // CDNA wavefront — 64 lanes, not 32

LO_DEVICE float tofloat(bf16 v) {
  union { unsigned int u; float f; } c;
  c.u = ((unsigned int)*(unsigned short*)&v) << 16;
  return c.f;
}

LO_DEVICE bf16 tobf16(float f) {
  // round-to-nearest-even f32 -> bf16
  union { float f; unsigned int u; } c; c.f = f;
  unsigned int lsb = (c.u >> 16) & 1;
  unsigned int rounded = c.u + 0x7fffu + lsb;
  unsigned short h = (unsigned short)(rounded >> 16);
  if ((c.u & 0x7f800000u) == 0x7f800000u) h = (unsigned short)(c.u >> 16); // inf/nan passthrough
  return *(bf16*)&h;
}

// ceil-div
constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

// Bijective XCD-aware block remap (guide §5: 8 XCDs, private L2s; consecutive
// blocks should land on ONE XCD to share operand panels in its L2).
LO_DEVICE int xcd_swizzle(int bid, int nblocks) {
  constexpr int NXCD = 8;
  if (nblocks < 2 * NXCD) return bid;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int q = nblocks / NXCD, r = nblocks % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// ---- fast integer division (magic numbers) --------------------------------
// q = n / d exact for n < 2^31: mg==0 -> d==1; mg==1 -> d==2^s (q = n>>s);
// else q = (n*mg) >> (32+s) with mg = ceil(2^(31+s0)/d), s = s0-1 (the
// 2^(32+s) formulation overflows u32 — r1 hard-won invariant).
struct FDiv { unsigned mg = 0; int s = 0; };

LO_DEVICE unsigned fdiv2(unsigned n, FDiv f) {
  if (f.mg == 0) return n;
  if (f.mg == 1) return n >> f.s;
  return (unsigned)(((unsigned long long)n * f.mg) >> 32) >> f.s;
}

inline void mkmagic(unsigned d, FDiv& f) {
  if (d <= 1) { f.mg = 0; f.s = 0; return; }
  int s = 0;
  while ((1u << s) < d) ++s;
  if ((1u << s) == d) { f.mg = 1; f.s = s; return; }
  const unsigned long long L = 1ull << (31 + s);
  f.mg = (unsigned)((L + d - 1) / d);
  f.s = s - 1;
}

}  // namespace lo
