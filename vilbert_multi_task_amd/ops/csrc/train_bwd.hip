// Training-path fused kernels: LayerNorm forward-with-stats + backward, and
// bias+GELU backward. These replace the PyTorch autograd kernel chains that
// profiles/r05_training_step.md measured at ~25% of the training step
// (elementwise adds, LN grad_input + PartGradGammaBeta, gelu_backward,
// bias-grad reduce):
//   - forward fuses the residual add into the LN pass (torch does a
//     separate add kernel: one extra read+write of the full activation),
//   - LN backward is 2 kernels instead of 3 (grad_input one-wave-per-row;
//     param grads via a fixed-chunk partial buffer, summed deterministically
//     with a torch reduce — no atomics, bitwise-reproducible),
//   - GELU backward emits grad_pre AND the bias-grad partials in ONE pass
//     (torch: gelu_backward + a separate sum-over-rows reduce).
// All activations bf16, statistics and parameter-gradient accumulation f32.
// Reference parity: backward of LN(x+res)*w+b and of gelu(x@W^T+b) exactly as
// autograd computes them (tests/test_gpu_ops.py compares against a full
// fp32 torch.autograd reference).

#include "common.h"

#define LNB_MAX_CHUNKS 4  // 4*8*64 = dims up to 2048 cached in registers

namespace {

DEV void load8(const bf16* p, float* out) {
  const uint4 raw = *reinterpret_cast<const uint4*>(p);
  const unsigned int w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    out[2 * i] = us2f((unsigned short)(w[i] & 0xffff));
    out[2 * i + 1] = us2f((unsigned short)(w[i] >> 16));
  }
}
DEV void store8(bf16* p, const float* in) {
  uint4 raw;
  unsigned int w[4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    w[i] = (unsigned int)f2us(in[2 * i]) | ((unsigned int)f2us(in[2 * i + 1]) << 16);
  raw.x = w[0]; raw.y = w[1]; raw.z = w[2]; raw.w = w[3];
  *reinterpret_cast<uint4*>(p) = raw;
}

// d/dv gelu(v) with exact-erf gelu (matches the forward in elementwise.hip)
DEV float dgelu(float v) {
  const float kInvSqrt2 = 0.70710678118654752f;
  const float kInvSqrt2Pi = 0.39894228040143268f;
  const float cdf = 0.5f * (1.0f + erff(v * kInvSqrt2));
  return cdf + v * kInvSqrt2Pi * __expf(-0.5f * v * v);
}

}  // namespace

// ---------------------------------------------------------------------------
// LN forward with saved stats.  y = LN(x (+res)) * w + b ; also writes
// xs = x+res (bf16, only when HAS_RES — otherwise the caller reuses x),
// mean/rstd (f32 per row).  One wave per row, grid-stride.
// ---------------------------------------------------------------------------
template <bool HAS_RES>
__global__ void ln_fwd_train_kernel(const bf16* __restrict__ x,
                                    const bf16* __restrict__ res,
                                    const bf16* __restrict__ w,
                                    const bf16* __restrict__ b,
                                    bf16* __restrict__ y,
                                    bf16* __restrict__ xs_out,
                                    float* __restrict__ mean_out,
                                    float* __restrict__ rstd_out,
                                    long rows, int dim, float eps) {
  const int lane = lane_id();
  const int wid = wave_id();
  const int waves_per_blk = blockDim.x / WAVE;
  const bool vec8 = (dim % 8) == 0;
  const int nchunk = vec8 ? ceil_div(dim / 8, WAVE) : 0;
  const bool cached = vec8 && nchunk <= LNB_MAX_CHUNKS;
  float cache[LNB_MAX_CHUNKS * 8];

  for (long row = (long)blockIdx.x * waves_per_blk + wid; row < rows;
       row += (long)gridDim.x * waves_per_blk) {
    const bf16* xr = x + row * dim;
    const bf16* rr = HAS_RES ? res + row * dim : nullptr;
    float s = 0.f, sq = 0.f;
    if (cached) {
#pragma unroll
      for (int c = 0; c < LNB_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          load8(xr + j, &cache[c * 8]);
          if (HAS_RES) {
            float r8[8];
            load8(rr + j, r8);
#pragma unroll
            for (int i = 0; i < 8; ++i) cache[c * 8 + i] += r8[i];
            store8(xs_out + row * dim + j, &cache[c * 8]);
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
        float v = bf2f(xr[j]);
        if (HAS_RES) {
          v += bf2f(rr[j]);
          xs_out[row * dim + j] = f2bf(v);
        }
        s += v;
        sq += v * v;
      }
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    const float mean = s / dim;
    const float var = sq / dim - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    if (cached) {
#pragma unroll
      for (int c = 0; c < LNB_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float w8[8], b8[8], o8[8];
          load8(w + j, w8);
          load8(b + j, b8);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            o8[i] = (cache[c * 8 + i] - mean) * rstd * w8[i] + b8[i];
          store8(y + row * dim + j, o8);
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        // NOTE: when HAS_RES the summed value was just written to xs_out;
        // re-read x/res instead of xs_out to avoid the store-load ordering.
        float v = bf2f(xr[j]);
        if (HAS_RES) v += bf2f(rr[j]);
        y[row * dim + j] = f2bf((v - mean) * rstd * bf2f(w[j]) + bf2f(b[j]));
      }
    }
  }
}

// ---------------------------------------------------------------------------
// LN backward, input side.  One wave per row:
//   g      = grad_y * w
//   xhat   = (xs - mean) * rstd
//   gx     = (g - xhat * mean(g*xhat) - mean(g)) * rstd
// ---------------------------------------------------------------------------
__global__ void ln_bwd_input_kernel(const bf16* __restrict__ grad_y,
                                    const bf16* __restrict__ xs,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const bf16* __restrict__ w,
                                    bf16* __restrict__ grad_xs,
                                    long rows, int dim) {
  const int lane = lane_id();
  const int wid = wave_id();
  const int waves_per_blk = blockDim.x / WAVE;
  const bool vec8 = (dim % 8) == 0;
  const int nchunk = vec8 ? ceil_div(dim / 8, WAVE) : 0;
  const bool cached = vec8 && nchunk <= LNB_MAX_CHUNKS;
  float gc[LNB_MAX_CHUNKS * 8];   // g = grad_y * w
  float xc[LNB_MAX_CHUNKS * 8];   // xhat

  for (long row = (long)blockIdx.x * waves_per_blk + wid; row < rows;
       row += (long)gridDim.x * waves_per_blk) {
    const float mu = mean[row], rs = rstd[row];
    const bf16* gyr = grad_y + row * dim;
    const bf16* xsr = xs + row * dim;
    float s1 = 0.f, s2 = 0.f;  // sum(g*xhat), sum(g)
    if (cached) {
#pragma unroll
      for (int c = 0; c < LNB_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float gy8[8], xs8[8], w8[8];
          load8(gyr + j, gy8);
          load8(xsr + j, xs8);
          load8(w + j, w8);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            const float g = gy8[i] * w8[i];
            const float xh = (xs8[i] - mu) * rs;
            gc[c * 8 + i] = g;
            xc[c * 8 + i] = xh;
            s1 += g * xh;
            s2 += g;
          }
        }
      }
      s1 = wave_sum(s1) / dim;
      s2 = wave_sum(s2) / dim;
#pragma unroll
      for (int c = 0; c < LNB_MAX_CHUNKS; ++c) {
        if (c >= nchunk) break;
        const int j = (c * WAVE + lane) * 8;
        if (j < dim) {
          float o8[8];
#pragma unroll
          for (int i = 0; i < 8; ++i)
            o8[i] = (gc[c * 8 + i] - xc[c * 8 + i] * s1 - s2) * rs;
          store8(grad_xs + row * dim + j, o8);
        }
      }
    } else {
      for (int j = lane; j < dim; j += WAVE) {
        const float g = bf2f(gyr[j]) * bf2f(w[j]);
        const float xh = (bf2f(xsr[j]) - mu) * rs;
        s1 += g * xh;
        s2 += g;
      }
      s1 = wave_sum(s1) / dim;
      s2 = wave_sum(s2) / dim;
      for (int j = lane; j < dim; j += WAVE) {
        const float g = bf2f(gyr[j]) * bf2f(w[j]);
        const float xh = (bf2f(xsr[j]) - mu) * rs;
        grad_xs[row * dim + j] = f2bf((g - xh * s1 - s2) * rs);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// LN backward, parameter side.  Column-parallel partial sums into a fixed
// [nchunks, dim] f32 buffer (deterministic: fixed chunk count, fixed order,
// final sum is one torch reduce).  grad_w[j] = sum_r gy*xhat, grad_b[j] = sum_r gy.
// One thread per column; blockIdx.y picks the row chunk.
// ---------------------------------------------------------------------------
__global__ void ln_bwd_param_kernel(const bf16* __restrict__ grad_y,
                                    const bf16* __restrict__ xs,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ gw_part,
                                    float* __restrict__ gb_part,
                                    long rows, int dim, int rows_per_chunk) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= dim) return;
  const long r0 = (long)blockIdx.y * rows_per_chunk;
  const long r1 = min(rows, r0 + rows_per_chunk);
  float gw = 0.f, gb = 0.f;
  for (long r = r0; r < r1; ++r) {
    const float gy = bf2f(grad_y[r * dim + j]);
    gw += gy * (bf2f(xs[r * dim + j]) - mean[r]) * rstd[r];
    gb += gy;
  }
  gw_part[(long)blockIdx.y * dim + j] = gw;
  gb_part[(long)blockIdx.y * dim + j] = gb;
}

// ---------------------------------------------------------------------------
// bias+GELU backward:  grad_pre = grad_y * dgelu(pre)  and the bias-grad
// partials in the same pass (grad_b = sum_rows grad_pre).  Column-parallel
// like ln_bwd_param: thread j loops its chunk's rows (coalesced across the
// 256 adjacent columns of the block).
// ---------------------------------------------------------------------------
__global__ void bias_gelu_bwd_kernel(const bf16* __restrict__ grad_y,
                                     const bf16* __restrict__ pre,
                                     bf16* __restrict__ grad_pre,
                                     float* __restrict__ gb_part,
                                     long rows, int dim, int rows_per_chunk) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= dim) return;
  const long r0 = (long)blockIdx.y * rows_per_chunk;
  const long r1 = min(rows, r0 + rows_per_chunk);
  float gb = 0.f;
  for (long r = r0; r < r1; ++r) {
    const float gp = bf2f(grad_y[r * dim + j]) * dgelu(bf2f(pre[r * dim + j]));
    grad_pre[r * dim + j] = f2bf(gp);
    gb += gp;
  }
  gb_part[(long)blockIdx.y * dim + j] = gb;
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
static int row_grid(long rows, int waves_per_blk) {
  long g = (rows + waves_per_blk - 1) / waves_per_blk;
  return (int)min(g, (long)65535);
}

void launch_ln_fwd_train(const bf16* x, const bf16* res, const bf16* w,
                         const bf16* b, bf16* y, bf16* xs, float* mean,
                         float* rstd, long rows, int dim, float eps,
                         hipStream_t stream) {
  dim3 block(256);
  dim3 grid(row_grid(rows, 4));
  if (res)
    hipLaunchKernelGGL((ln_fwd_train_kernel<true>), grid, block, 0, stream, x,
                       res, w, b, y, xs, mean, rstd, rows, dim, eps);
  else
    hipLaunchKernelGGL((ln_fwd_train_kernel<false>), grid, block, 0, stream, x,
                       res, w, b, y, xs, mean, rstd, rows, dim, eps);
}

void launch_ln_bwd_input(const bf16* gy, const bf16* xs, const float* mean,
                         const float* rstd, const bf16* w, bf16* gxs,
                         long rows, int dim, hipStream_t stream) {
  dim3 block(256);
  dim3 grid(row_grid(rows, 4));
  hipLaunchKernelGGL(ln_bwd_input_kernel, grid, block, 0, stream, gy, xs, mean,
                     rstd, w, gxs, rows, dim);
}

void launch_ln_bwd_param(const bf16* gy, const bf16* xs, const float* mean,
                         const float* rstd, float* gw_part, float* gb_part,
                         long rows, int dim, int nchunks, int rows_per_chunk,
                         hipStream_t stream) {
  dim3 block(256);
  dim3 grid(ceil_div(dim, 256), nchunks);
  hipLaunchKernelGGL(ln_bwd_param_kernel, grid, block, 0, stream, gy, xs, mean,
                     rstd, gw_part, gb_part, rows, dim, rows_per_chunk);
}

void launch_bias_gelu_bwd(const bf16* gy, const bf16* pre, bf16* gpre,
                          float* gb_part, long rows, int dim, int nchunks,
                          int rows_per_chunk, hipStream_t stream) {
  dim3 block(256);
  dim3 grid(ceil_div(dim, 256), nchunks);
  hipLaunchKernelGGL(bias_gelu_bwd_kernel, grid, block, 0, stream, gy, pre,
                     gpre, gb_part, rows, dim, rows_per_chunk);
}
