// FP8 (OCP e4m3) producer-side quantization support.
//
// The standalone-quantization fp8 mode measured SLOWER than bf16 (amax +
// cast passes per GEMM — docs/PERFORMANCE.md). These pieces make
// quantization ~free with DELAYED SCALING: producing kernels emit an e4m3
// copy scaled by the PREVIOUS step's scale while atomically accumulating
// this step's amax; one tiny kernel per step refreshes every site's scale.
// All of it is hipGraph-capturable (no host syncs).
//
// gfx950 fp8 is OCP e4m3fn (NOT MI300X fnuz — cdna_hip_programming.md §4);
// __hip_fp8_e4m3 converts with the hardware cvt ops.

#include <hip/hip_fp8.h>

#include "common.h"

#define E4M3_MAX 448.0f

DEV void atomic_max_f32_nonneg(float* addr, float v) {
  // monotone uint ordering holds for non-negative floats; read-guard skips
  // the atomic when the global amax already covers v (contention fix)
  if (v <= *reinterpret_cast<volatile float*>(addr)) return;
  atomicMax(reinterpret_cast<unsigned int*>(addr), __float_as_uint(v));
}

DEV unsigned char f2e4m3(float v) {
  __hip_fp8_e4m3 q(v);
  return q.__x;
}

DEV uint2 pack8_e4m3_q(const float* v, float inv) {
  float c[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    c[i] = fminf(fmaxf(v[i] * inv, -E4M3_MAX), E4M3_MAX);
  int lo = 0, hi = 0;
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(c[0], c[1], lo, false);
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(c[2], c[3], lo, true);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(c[4], c[5], hi, false);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(c[6], c[7], hi, true);
  uint2 r;
  r.x = (unsigned int)lo;
  r.y = (unsigned int)hi;
  return r;
}

DEV void VecIO_q_load(const bf16* p, float* out) {
  const uint4 raw = *reinterpret_cast<const uint4*>(p);
  const unsigned int w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    out[2 * i] = us2f((unsigned short)(w[i] & 0xffff));
    out[2 * i + 1] = us2f((unsigned short)(w[i] >> 16));
  }
}

// ---------------------------------------------------------------------------
// standalone quantize: y8 = clamp(x / scale[site]); amax[site] accumulates
// ---------------------------------------------------------------------------

__global__ void quantize_fp8_kernel(const bf16* __restrict__ x,
                                    unsigned char* __restrict__ y8,
                                    const float* __restrict__ scales,
                                    float* __restrict__ amaxes, int site,
                                    long n) {
  const float inv = 1.0f / scales[site];
  float local = 0.f;
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      float v[8];
      VecIO_q_load(x + i, v);
#pragma unroll
      for (int j = 0; j < 8; ++j) local = fmaxf(local, fabsf(v[j]));
      *reinterpret_cast<uint2*>(y8 + i) = pack8_e4m3_q(v, inv);
    } else {
      for (long k = i; k < n; ++k) {
        const float v = bf2f(x[k]);
        local = fmaxf(local, fabsf(v));
        y8[k] = f2e4m3(fminf(fmaxf(v * inv, -E4M3_MAX), E4M3_MAX));
      }
    }
  }
  local = wave_max(local);
  if (lane_id() == 0 && local > 0.f)
    atomic_max_f32_nonneg(&amaxes[site], local);
}

// ---------------------------------------------------------------------------
// per-step scale refresh for ALL sites: scale = max(amax, eps)/448; amax = 0
// ---------------------------------------------------------------------------

__global__ void update_fp8_scales_kernel(float* __restrict__ scales,
                                         float* __restrict__ inv_scales,
                                         float* __restrict__ amaxes,
                                         int nsites) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nsites) {
    const float a = amaxes[i];
    if (a > 0.f) {
      const float s = fmaxf(a, 1e-8f) / E4M3_MAX;
      scales[i] = s;
      inv_scales[i] = 1.0f / s;  // hipBLASLt D_SCALE multiplies before cast
    }
    amaxes[i] = 0.f;
  }
}

void launch_quantize_fp8(const bf16* x, unsigned char* y8, const float* scales,
                         float* amaxes, int site, long n, hipStream_t stream) {
  const int block = 256;
  long want = (n + 8L * block - 1) / (8L * block);
  const int grid = (int)(want < 1024 ? want : 1024);
  hipLaunchKernelGGL(quantize_fp8_kernel, dim3(grid), dim3(block), 0, stream, x,
                     y8, scales, amaxes, site, n);
}

void launch_update_fp8_scales(float* scales, float* inv_scales, float* amaxes,
                              int nsites, hipStream_t stream) {
  hipLaunchKernelGGL(update_fp8_scales_kernel, dim3(ceil_div(nsites, 64)),
                     dim3(64), 0, stream, scales, inv_scales, amaxes, nsites);
}
