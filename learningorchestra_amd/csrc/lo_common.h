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

}  // namespace lo
