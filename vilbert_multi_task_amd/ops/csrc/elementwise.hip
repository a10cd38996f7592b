// Fused memory-bound kernels: residual+LayerNorm, bias+GELU, embedding+LN.
//
// These replace the implicit CUDA elementwise/normalization kernels of the
// reference's PyTorch forward (SURVEY.md §2.3 "Implicit CUDA kernels") with
// hand-written CDNA4 code: one wave per row, bf16 loads vectorized 8-wide
// (16 B/lane — cdna_hip_programming.md Guideline 13: scalar bf16 loads are
// ~2-2.5x slower), f32 accumulation, fused residual add so the tensor is
// read once from HBM instead of twice.

#include <hip/hip_fp8.h>

#include "common.h"
#include <cstdlib>

#define E4M3_MAX_E 448.0f

DEV void atomic_max_f32_nonneg_e(float* addr, float v) {
  // read-guard: skip the atomic once the global amax already covers v —
  // without it every wave serializes on one L2 dword (measured: the FP8OUT
  // LN was 2.8x slower than plain LN at unchanged occupancy/conversions)
  if (v <= *reinterpret_cast<volatile float*>(addr)) return;
  atomicMax(reinterpret_cast<unsigned int*>(addr), __float_as_uint(v));
}
DEV unsigned char f2e4m3_e(float v) {
  __hip_fp8_e4m3 q(fminf(fmaxf(v, -E4M3_MAX_E), E4M3_MAX_E));
  return q.__x;
}
// hardware packed convert: 8 floats -> 8 e4m3 bytes in 4 v_cvt_pk_fp8_f32
// (the __hip_fp8_e4m3 constructor is a software path — it made the FP8OUT
// LayerNorm 2.8x slower than plain; profiles/prof_fp8)
DEV uint2 pack8_e4m3(const float* v, float inv) {
  float c[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    c[i] = fminf(fmaxf(v[i] * inv, -E4M3_MAX_E), E4M3_MAX_E);
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

// ---------------------------------------------------------------------------
// residual + LayerNorm.  One wave per row; rows assigned grid-stride.
// dim % 8 == 0 fast path (uint4 = 8 bf16); scalar fallback otherwise.
// ---------------------------------------------------------------------------

template <typename T>
struct VecIO;

template <>
struct VecIO<bf16> {
  // 8 elements = 16 bytes
  DEV static void load8(const bf16* p, float* out) {
    const uint4 raw = *reinterpret_cast<const uint4*>(p);
    const unsigned int w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      out[2 * i] = us2f((unsigned short)(w[i] & 0xffff));
      out[2 * i + 1] = us2f((unsigned short)(w[i] >> 16));
    }
  }
  DEV static void store8(bf16* p, const float* in) {
    uint4 raw;
    unsigned int w[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      w[i] = (unsigned int)f2us(in[2 * i]) | ((unsigned int)f2us(in[2 * i + 1]) << 16);
    }
    raw.x = w[0]; raw.y = w[1]; raw.z = w[2]; raw.w = w[3];
    *reinterpret_cast<uint4*>(p) = raw;
  }
  DEV static float ld(const bf16* p) { return bf2f(*p); }
  DEV static void st(bf16* p, float v) { *p = f2bf(v); }
};

template <>
struct VecIO<float> {
  DEV static void load8(const float* p, float* out) {
    const float4 a = *reinterpret_cast<const float4*>(p);
    const float4 b = *reinterpret_cast<const float4*>(p + 4);
    out[0] = a.x; out[1] = a.y; out[2] = a.z; out[3] = a.w;
    out[4] = b.x; out[5] = b.y; out[6] = b.z; out[7] = b.w;
  }
  DEV static void store8(float* p, const float* in) {
    *reinterpret_cast<float4*>(p) = make_float4(in[0], in[1], in[2], in[3]);
    *reinterpret_cast<float4*>(p + 4) = make_float4(in[4], in[5], in[6], in[7]);
  }
  DEV static float ld(const float* p) { return *p; }
  DEV static void st(float* p, float v) { *p = v; }
};

// MAX_CHUNKS*8*64 = max dim cached in registers per wave (4 -> 2048; dims
// beyond that re-read from L2 in the normalize pass).
#define LN_MAX_CHUNKS 4

template <typename T, bool HAS_RES, bool FP8OUT = false>
__global__ void residual_ln_kernel(const T* __restrict__ x,
                                   const T* __restrict__ res,
                                   const T* __restrict__ w,
                                   const T* __restrict__ b,
                                   T* __restrict__ y,
                                   long rows, int dim, float eps,
                                   unsigned char* __restrict__ y8 = nullptr,
                                   const float* __restrict__ scales = nullptr,
                                   float* __restrict__ amaxes = nullptr,
                                   int site = 0) {
  // FP8OUT: also emit an e4m3 copy of y scaled by scales[site] (delayed
  // scaling: previous step's scale) and accumulate this step's amax —
  // makes fp8 GEMM-input quantization free (docs/ROADMAP.md item 1).
  const float fp8_inv = FP8OUT ? 1.0f / scales[site] : 0.f;
  float fp8_amax = 0.f;
  const int lane = lane_id();
  const int wid = wave_id();
  const int waves_per_blk = blockDim.x / WAVE;
  const bool vec8 = (dim % 8) == 0;
  const int nchunk = vec8 ? ceil_div(dim / 8, WAVE) : 0;
  const bool cached = vec8 && nchunk <= LN_MAX_CHUNKS;
  float cache[LN_MAX_CHUNKS * 8];

  for (long row = (long)blockIdx.x * waves_per_blk + wid; row < rows;
       row += (long)gridDim.x * waves_per_blk) {
    const T* xr = x + row * dim;
    const T* rr = HAS_RES ? res + row * dim : nullptr;
    T* yr = y + row * dim;
    float s = 0.f, sq = 0.f;
    if (cached) {
#pragma unroll
      for (int c = 0; c < LN_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          VecIO<T>::load8(xr + j, &cache[c * 8]);
          if (HAS_RES) {
            float r8[8];
            VecIO<T>::load8(rr + j, r8);
#pragma unroll
            for (int i = 0; i < 8; ++i) cache[c * 8 + i] += r8[i];
          }
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            s += cache[c * 8 + i];
            sq += cache[c * 8 + i] * cache[c * 8 + i];
          }
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        float v = VecIO<T>::ld(xr + j);
        if (HAS_RES) v += VecIO<T>::ld(rr + j);
        s += v;
        sq += v * v;
      }
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    const float mean = s / dim;
    const float var = sq / dim - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (cached) {
#pragma unroll
      for (int c = 0; c < LN_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float w8[8], b8[8], o8[8];
          VecIO<T>::load8(w + j, w8);
          VecIO<T>::load8(b + j, b8);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            o8[i] = (cache[c * 8 + i] - mean) * rstd * w8[i] + b8[i];
          VecIO<T>::store8(yr + j, o8);
          if (FP8OUT) {
#pragma unroll
            for (int i = 0; i < 8; ++i)
              fp8_amax = fmaxf(fp8_amax, fabsf(o8[i]));
            *reinterpret_cast<uint2*>(y8 + row * dim + j) =
                pack8_e4m3(o8, fp8_inv);
          }
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        float v = VecIO<T>::ld(xr + j);
        if (HAS_RES) v += VecIO<T>::ld(rr + j);
        const float o = (v - mean) * rstd * VecIO<T>::ld(w + j) + VecIO<T>::ld(b + j);
        VecIO<T>::st(yr + j, o);
        if (FP8OUT) {
          fp8_amax = fmaxf(fp8_amax, fabsf(o));
          y8[row * dim + j] = f2e4m3_e(o * fp8_inv);
        }
      }
    }
  }
  if (FP8OUT) {
    fp8_amax = wave_max(fp8_amax);
    if (lane_id() == 0 && fp8_amax > 0.f)
      atomic_max_f32_nonneg_e(&amaxes[site], fp8_amax);
  }
}

// ---------------------------------------------------------------------------
// bias + exact GELU:  y = 0.5*(x+b)*(1+erf((x+b)/sqrt(2)))
// ---------------------------------------------------------------------------

template <typename T, bool HAS_BIAS, bool FP8OUT = false>
__global__ void bias_gelu_kernel(const T* __restrict__ x,
                                 const T* __restrict__ bias,
                                 T* __restrict__ y,
                                 long n, int dim,
                                 unsigned char* __restrict__ y8 = nullptr,
                                 const float* __restrict__ scales = nullptr,
                                 float* __restrict__ amaxes = nullptr,
                                 int site = 0) {
  const float fp8_inv = FP8OUT ? 1.0f / scales[site] : 0.f;
  float fp8_amax = 0.f;
  const long i8 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long stride8 = (long)gridDim.x * blockDim.x * 8;
  constexpr float kInvSqrt2 = 0.70710678118654752440f;
  for (long i = i8; i < n; i += stride8) {
    if (i + 8 <= n && (!HAS_BIAS || ((i % dim) + 8 <= dim))) {
      float v[8];
      VecIO<T>::load8(x + i, v);
      if (HAS_BIAS) {
        float b8[8];
        VecIO<T>::load8(bias + (i % dim), b8);
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] += b8[j];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v[j] = 0.5f * v[j] * (1.0f + erff(v[j] * kInvSqrt2));
      VecIO<T>::store8(y + i, v);
      if (FP8OUT) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          fp8_amax = fmaxf(fp8_amax, fabsf(v[j]));
        *reinterpret_cast<uint2*>(y8 + i) = pack8_e4m3(v, fp8_inv);
      }
    } else {
      for (long k = i; k < min(i + 8, n); ++k) {
        float v = VecIO<T>::ld(x + k);
        if (HAS_BIAS) v += VecIO<T>::ld(bias + (k % dim));
        const float g = 0.5f * v * (1.0f + erff(v * kInvSqrt2));
        VecIO<T>::st(y + k, g);
        if (FP8OUT) {
          fp8_amax = fmaxf(fp8_amax, fabsf(g));
          y8[k] = f2e4m3_e(g * fp8_inv);
        }
      }
    }
  }
  if (FP8OUT) {
    fp8_amax = wave_max(fp8_amax);
    if (lane_id() == 0 && fp8_amax > 0.f)
      atomic_max_f32_nonneg_e(&amaxes[site], fp8_amax);
  }
}

// ---------------------------------------------------------------------------
// word + position + segment embedding gather, sum, LayerNorm.
// One wave per token (replaces three gathers + add + LN round trips).
// ---------------------------------------------------------------------------

template <typename T>
__global__ void embedding_ln_kernel(const long* __restrict__ ids,
                                    const long* __restrict__ pos_ids,
                                    const long* __restrict__ type_ids,
                                    const T* __restrict__ word_w,
                                    const T* __restrict__ pos_w,
                                    const T* __restrict__ type_w,
                                    const T* __restrict__ ln_w,
                                    const T* __restrict__ ln_b,
                                    T* __restrict__ y,
                                    long tokens, int dim, float eps) {
  const int lane = lane_id();
  const int wid = wave_id();
  const int waves_per_blk = blockDim.x / WAVE;
  const bool vec8 = (dim % 8) == 0;
  const int nchunk = vec8 ? ceil_div(dim / 8, WAVE) : 0;
  const bool cached = vec8 && nchunk <= LN_MAX_CHUNKS;
  float cache[LN_MAX_CHUNKS * 8];

  for (long tok = (long)blockIdx.x * waves_per_blk + wid; tok < tokens;
       tok += (long)gridDim.x * waves_per_blk) {
    const T* wrow = word_w + (long)ids[tok] * dim;
    const T* prow = pos_w + (long)pos_ids[tok] * dim;
    const T* trow = type_w + (long)type_ids[tok] * dim;
    T* yr = y + tok * dim;
    float s = 0.f, sq = 0.f;
    if (cached) {
#pragma unroll
      for (int c = 0; c < LN_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float a[8], b8[8], c8[8];
          VecIO<T>::load8(wrow + j, a);
          VecIO<T>::load8(prow + j, b8);
          VecIO<T>::load8(trow + j, c8);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            cache[c * 8 + i] = a[i] + b8[i] + c8[i];
            s += cache[c * 8 + i];
            sq += cache[c * 8 + i] * cache[c * 8 + i];
          }
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        float v = VecIO<T>::ld(wrow + j) + VecIO<T>::ld(prow + j) + VecIO<T>::ld(trow + j);
        s += v;
        sq += v * v;
      }
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    const float mean = s / dim;
    const float rstd = rsqrtf(sq / dim - mean * mean + eps);
    if (cached) {
#pragma unroll
      for (int c = 0; c < LN_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float w8[8], b8[8], o8[8];
          VecIO<T>::load8(ln_w + j, w8);
          VecIO<T>::load8(ln_b + j, b8);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            o8[i] = (cache[c * 8 + i] - mean) * rstd * w8[i] + b8[i];
          VecIO<T>::store8(yr + j, o8);
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        float v = VecIO<T>::ld(wrow + j) + VecIO<T>::ld(prow + j) + VecIO<T>::ld(trow + j);
        VecIO<T>::st(yr + j, (v - mean) * rstd * VecIO<T>::ld(ln_w + j) + VecIO<T>::ld(ln_b + j));
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Software-pipelined residual+LN for the serving dims (dim % 8 == 0,
// dim <= 1024): the baseline wave-per-row kernel measures 3.8-4.3 TB/s of
// the 8 TB/s roofline (profiles/r09) because 124 VGPRs cap it at 16
// waves/CU and each row's load -> reduce -> normalize chain runs serially
// per wave: ~48 KB in flight per CU sits exactly at the bandwidth-latency
// product. Here each wave ping-pongs two RAW uint4 row buffers (no f32
// cache) and issues row n+1's loads before reducing row n, so HBM reads fly
// under the reduction+store of the previous row; w/gamma and b/beta are
// hoisted into registers once per wave. bf16 only; fp8 stays on the
// baseline kernel.
template <bool HAS_RES, bool FP8OUT = false>
__global__ void residual_ln_pipe_kernel(const bf16* __restrict__ x,
                                        const bf16* __restrict__ res,
                                        const bf16* __restrict__ w,
                                        const bf16* __restrict__ b,
                                        bf16* __restrict__ y,
                                        long rows, int dim, float eps,
                                        unsigned char* __restrict__ y8 = nullptr,
                                        const float* __restrict__ scales = nullptr,
                                        float* __restrict__ amaxes = nullptr,
                                        int site = 0) {
  const float fp8_inv = FP8OUT ? 1.0f / scales[site] : 0.f;
  float fp8_amax = 0.f;
  const int lane = lane_id();
  const int wid = wave_id();
  const int waves_per_blk = blockDim.x / WAVE;
  const int nchunk = ceil_div(dim / 8, WAVE);  // <= 2 (dim <= 1024)
  const long stride = (long)gridDim.x * waves_per_blk;

  // gamma/beta raw (L2-resident; kept as packed uint4 — hoisting them as
  // f32 arrays cost 32 VGPRs and spilled the HAS_RES variant)
  uint4 wraw[2], braw[2];
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int j = (c * WAVE + lane) * 8;
    if (c < nchunk && j < dim) {
      wraw[c] = *reinterpret_cast<const uint4*>(w + j);
      braw[c] = *reinterpret_cast<const uint4*>(b + j);
    }
  }

  struct Raw { uint4 xr[2], rr[2]; };
  auto load_raw = [&](long row, Raw& r) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int j = (c * WAVE + lane) * 8;
      if (c < nchunk && j < dim) {
        r.xr[c] = *reinterpret_cast<const uint4*>(x + row * dim + j);
        if (HAS_RES)
          r.rr[c] = *reinterpret_cast<const uint4*>(res + row * dim + j);
      }
    }
  };
  auto cvt8 = [&](const uint4& a, float* out) {
    const unsigned int wds[4] = {a.x, a.y, a.z, a.w};
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      out[2 * i] = us2f((unsigned short)(wds[i] & 0xffff));
      out[2 * i + 1] = us2f((unsigned short)(wds[i] >> 16));
    }
  };
  auto process = [&](long row, const Raw& r) {
    float s = 0.f, sq = 0.f;
    float v8[2][8];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int j = (c * WAVE + lane) * 8;
      if (c < nchunk && j < dim) {
        cvt8(r.xr[c], v8[c]);
        if (HAS_RES) {
          float t8[8];
          cvt8(r.rr[c], t8);
#pragma unroll
          for (int i = 0; i < 8; ++i) v8[c][i] += t8[i];
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          s += v8[c][i];
          sq += v8[c][i] * v8[c][i];
        }
      }
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    const float mean = s / dim;
    const float rstd = rsqrtf(sq / dim - mean * mean + eps);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int j = (c * WAVE + lane) * 8;
      if (c < nchunk && j < dim) {
        float o8[8], w8[8], b8[8];
        cvt8(wraw[c], w8);
        cvt8(braw[c], b8);
#pragma unroll
        for (int i = 0; i < 8; ++i)
          o8[i] = (v8[c][i] - mean) * rstd * w8[i] + b8[i];
        VecIO<bf16>::store8(y + row * dim + j, o8);
        if (FP8OUT) {
#pragma unroll
          for (int i = 0; i < 8; ++i)
            fp8_amax = fmaxf(fp8_amax, fabsf(o8[i]));
          *reinterpret_cast<uint2*>(y8 + row * dim + j) =
              pack8_e4m3(o8, fp8_inv);
        }
      }
    }
  };

  long row = (long)blockIdx.x * waves_per_blk + wid;
  if (row >= rows) goto done;
  Raw bufA, bufB;
  load_raw(row, bufA);
  while (true) {
    if (row + stride < rows) load_raw(row + stride, bufB);
    process(row, bufA);
    row += stride;
    if (row >= rows) break;
    if (row + stride < rows) load_raw(row + stride, bufA);
    process(row, bufB);
    row += stride;
    if (row >= rows) break;
  }
done:
  if (FP8OUT) {
    fp8_amax = wave_max(fp8_amax);
    if (lane_id() == 0 && fp8_amax > 0.f)
      atomic_max_f32_nonneg_e(&amaxes[site], fp8_amax);
  }
}

// ---------------------------------------------------------------------------
// host-side launchers (called from bindings.cpp)
// ---------------------------------------------------------------------------

void launch_residual_ln_fp8(const bf16* x, const bf16* res, const bf16* w,
                            const bf16* b, bf16* y, long rows, int dim,
                            float eps, unsigned char* y8, const float* scales,
                            float* amaxes, int site, hipStream_t stream) {
  const int block = 256;
  const int waves = block / WAVE;
  long want = (rows + waves - 1) / waves;
  const int grid = (int)(want < 2048 ? want : 2048);
  // measured: the pipe kernel's fp8 epilogue (pack + amax) costs more than
  // the load pipelining saves here (20.56k vs 20.72k q/s fp8 e2e) — the
  // fp8 path stays on the baseline kernel; VILBERT_LN_PIPE8=1 re-enables
  // the pipelined variant for iteration.
  static const bool pipe8 = [] {
    const char* e = getenv("VILBERT_LN_PIPE8");
    return e && e[0] == '1';
  }();
  if (pipe8 && dim % 8 == 0 && dim <= 1024) {
    if (res)
      hipLaunchKernelGGL((residual_ln_pipe_kernel<true, true>), dim3(grid),
                         dim3(block), 0, stream, x, res, w, b, y, rows, dim,
                         eps, y8, scales, amaxes, site);
    else
      hipLaunchKernelGGL((residual_ln_pipe_kernel<false, true>), dim3(grid),
                         dim3(block), 0, stream, x, res, w, b, y, rows, dim,
                         eps, y8, scales, amaxes, site);
    return;
  }
  if (res)
    hipLaunchKernelGGL((residual_ln_kernel<bf16, true, true>), dim3(grid),
                       dim3(block), 0, stream, x, res, w, b, y, rows, dim, eps,
                       y8, scales, amaxes, site);
  else
    hipLaunchKernelGGL((residual_ln_kernel<bf16, false, true>), dim3(grid),
                       dim3(block), 0, stream, x, res, w, b, y, rows, dim, eps,
                       y8, scales, amaxes, site);
}

template <typename T>
void launch_residual_ln(const T* x, const T* res, const T* w, const T* b, T* y,
                        long rows, int dim, float eps, hipStream_t stream) {
  const int block = 256;
  const int waves = block / WAVE;
  // grid cap: 2048 WGs = 8/CU (measured default); VILBERT_LN_GRID overrides
  // for occupancy probes
  static const long cap = [] {
    const char* e = getenv("VILBERT_LN_GRID");
    const long v = e ? atol(e) : 0;
    return v > 0 ? v : 2048L;
  }();
  const int grid = (int)min((rows + waves - 1) / waves, cap);
  static const bool pipe = [] {
    const char* e = getenv("VILBERT_LN_PIPE");
    return !(e && e[0] == '0');  // default on; 0 falls back to the baseline
  }();
  if (sizeof(T) == 2 && pipe && dim % 8 == 0 && dim <= 1024) {
    if (res)
      hipLaunchKernelGGL((residual_ln_pipe_kernel<true>), dim3(grid),
                         dim3(block), 0, stream, (const bf16*)x,
                         (const bf16*)res, (const bf16*)w, (const bf16*)b,
                         (bf16*)y, rows, dim, eps);
    else
      hipLaunchKernelGGL((residual_ln_pipe_kernel<false>), dim3(grid),
                         dim3(block), 0, stream, (const bf16*)x,
                         (const bf16*)res, (const bf16*)w, (const bf16*)b,
                         (bf16*)y, rows, dim, eps);
    return;
  }
  if (res)
    hipLaunchKernelGGL((residual_ln_kernel<T, true>), dim3(grid), dim3(block), 0,
                       stream, x, res, w, b, y, rows, dim, eps);
  else
    hipLaunchKernelGGL((residual_ln_kernel<T, false>), dim3(grid), dim3(block), 0,
                       stream, x, res, w, b, y, rows, dim, eps);
}

void launch_bias_gelu_fp8(const bf16* x, const bf16* bias, bf16* y, long n,
                          int dim, unsigned char* y8, const float* scales,
                          float* amaxes, int site, hipStream_t stream) {
  const int block = 256;
  const long want = (n + 8 * block - 1) / (8 * block);
  const int grid = (int)(want < 2048 ? want : 2048);
  if (bias)
    hipLaunchKernelGGL((bias_gelu_kernel<bf16, true, true>), dim3(grid),
                       dim3(block), 0, stream, x, bias, y, n, dim, y8, scales,
                       amaxes, site);
  else
    hipLaunchKernelGGL((bias_gelu_kernel<bf16, false, true>), dim3(grid),
                       dim3(block), 0, stream, x, bias, y, n, dim, y8, scales,
                       amaxes, site);
}

template <typename T>
void launch_bias_gelu(const T* x, const T* bias, T* y, long n, int dim,
                      hipStream_t stream) {
  const int block = 256;
  const long want = (n + 8 * block - 1) / (8 * block);
  const int grid = (int)min(want, (long)2048);
  if (bias)
    hipLaunchKernelGGL((bias_gelu_kernel<T, true>), dim3(grid), dim3(block), 0,
                       stream, x, bias, y, n, dim);
  else
    hipLaunchKernelGGL((bias_gelu_kernel<T, false>), dim3(grid), dim3(block), 0,
                       stream, x, bias, y, n, dim);
}

template <typename T>
void launch_embedding_ln(const long* ids, const long* pos_ids, const long* type_ids,
                         const T* word_w, const T* pos_w, const T* type_w,
                         const T* ln_w, const T* ln_b, T* y, long tokens, int dim,
                         float eps, hipStream_t stream) {
  const int block = 256;
  const int waves = block / WAVE;
  const int grid = (int)min((tokens + waves - 1) / waves, (long)2048);
  hipLaunchKernelGGL((embedding_ln_kernel<T>), dim3(grid), dim3(block), 0, stream,
                     ids, pos_ids, type_ids, word_w, pos_w, type_w, ln_w, ln_b, y,
                     tokens, dim, eps);
}

// explicit instantiations
template void launch_residual_ln<float>(const float*, const float*, const float*, const float*, float*, long, int, float, hipStream_t);
template void launch_residual_ln<bf16>(const bf16*, const bf16*, const bf16*, const bf16*, bf16*, long, int, float, hipStream_t);
template void launch_bias_gelu<float>(const float*, const float*, float*, long, int, hipStream_t);
template void launch_bias_gelu<bf16>(const bf16*, const bf16*, bf16*, long, int, hipStream_t);
template void launch_embedding_ln<float>(const long*, const long*, const long*, const float*, const float*, const float*, const float*, const float*, float*, long, int, float, hipStream_t);
template void launch_embedding_ln<bf16>(const long*, const long*, const long*, const bf16*, const bf16*, const bf16*, const bf16*, const bf16*, bf16*, long, int, float, hipStream_t);
