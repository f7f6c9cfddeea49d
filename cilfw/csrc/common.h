// cilfw CDNA4 kernel common helpers — gfx950 only, no dual paths.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) short bf16x8;   // MFMA A/B fragment
typedef __attribute__((ext_vector_type(4))) float f32x4;     // MFMA C/D fragment
typedef unsigned short bf16_t;                                // raw bf16 bits

DEV float bf2f(bf16_t v) {
  union { uint32_t u; float f; } x;
  x.u = ((uint32_t)v) << 16;
  return x.f;
}

DEV bf16_t f2bf(float f) {
  union { uint32_t u; float f; } x;
  x.f = f;
  uint32_t u = x.u;
  uint32_t r = (u + 0x7fffu + ((u >> 16) & 1u)) >> 16;  // RNE
  // quiet NaN passthrough
  if ((u & 0x7f800000u) == 0x7f800000u && (u & 0x7fffffu)) r = (u >> 16) | 1u;
  return (bf16_t)r;
}

DEV int cdiv_i(int a, int b) { return (a + b - 1) / b; }
static inline int cdiv(int a, int b) { return (a + b - 1) / b; }

// ---- MFMA 16x16x32 bf16 fragment maps (verified by tests/test_ops_gpu.py) ----
// A (16x32):  row i = lane & 15, k = (lane >> 4) * 8 + j   (j = 0..7)
// B (32x16):  col n = lane & 15, k = (lane >> 4) * 8 + j
// C/D (16x16): col = lane & 15, row = (lane >> 4) * 4 + reg
struct FragIdx {
  int half;   // lane & 15
  int quad;   // lane >> 4  (0..3)
};
DEV FragIdx frag_idx() {
  int lane = threadIdx.x & (WAVE - 1);
  return {lane & 15, lane >> 4};
}
