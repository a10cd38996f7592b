// Common device helpers for the gfx950 (CDNA4) ViLBERT kernels.
// Wave size is 64 on CDNA4 (cdna_hip_programming.md §1) — hard-coded.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

using bf16 = __hip_bfloat16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) short;  // 8 bf16 in 4 VGPRs
using short4v = __attribute__((ext_vector_type(4))) short;

DEV float bf2f(bf16 x) { return __bfloat162float(x); }
DEV bf16 f2bf(float x) { return __float2bfloat16(x); }

// bitcast an unsigned short holding bf16 bits to float
DEV float us2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}
DEV unsigned short f2us(float f) {
  union { unsigned int i; float f; } c;
  c.f = f;
  unsigned int i = c.i;
  // round-to-nearest-even bf16
  unsigned int lsb = (i >> 16) & 1;
  i += 0x7fffu + lsb;
  return (unsigned short)(i >> 16);
}

// full-wave (64-lane) butterfly reductions
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}
// 16-lane-group butterfly (groups = lanes sharing lane>>4) — used for the
// per-row reductions of the MFMA C-layout (col = lane&15).
DEV float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
DEV float group16_max(float v) {
#pragma unroll
  for (int off = 8; off; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

DEV int lane_id() { return threadIdx.x & (WAVE - 1); }
DEV int wave_id() { return threadIdx.x >> 6; }

__host__ __device__ static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }
