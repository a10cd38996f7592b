// Attention BACKWARD for the ViLBERT serving shapes (gfx950 MFMA) —
// training attention previously fell back to torch math.
//
// Forward (attention.hip, training variant) saves the NORMALIZED pre-dropout
// probs P [B,H,Lq,Lk] bf16 and the dropout keep-scale dm (0 or 1/(1-p));
// the context it produced is O = (dm .* P) @ V. Backward math:
//   dPt = dO @ V^T                      (dPt == d(P~) where P~ = dm .* P)
//   dP  = dm .* dPt
//   dS  = scale * P .* (dP - rowsum(dP .* P))
//   dQ  = dS @ K ; dK = dS^T @ Q ; dV = P~^T @ dO
//
// Split into two kernels so the LDS images fit (160 KiB/CU):
//  B1 (per (b,h), q-row stripes): stages K TRANSPOSED [D][LK_PAD] (the only
//     operand whose B-fragment is k-strided); dO rows and V rows feed the
//     dPt MFMAs straight from global (both are contiguous-K fragments);
//     softmax-backward runs on the dPt accumulator in registers with P/dm
//     preloaded batched (the r2 mask-load lesson: no per-element guarded
//     loads); dS goes to a per-wave row image for the dQ MFMAs AND to
//     global scratch bf16 for B2.
//  B2 (per (b,h), key stripes): stages TRANSPOSED images of Q, dO (from
//     global rows) and of dS, P~ (from the scratch/probs row layouts, pad
//     q-rows zeroed); computes dK = dS^T @ Q and dV = P~^T @ dO.
//
// Fragment maps as attention.hip. Parity: tests/test_gpu_ops.py compares
// dq/dk/dv against a full fp32 torch.autograd reference.

#include "common.h"

#define SWZB(row) (((((row) & 7) ^ (((row) >> 3) & 7))) << 4)
// row-length-masked form: the plain swizzle (up to 112 B) escapes a
// 64-byte row at *_PAD == 32 and collides across rows (same latent bug
// class as attention.hip's SWZR fix) — mask to the image's row length.
#define SWZBR(row, rbytes) (SWZB(row) & ((rbytes) - 1))

namespace {

DEV bf16x8 ld16(const bf16* p) {
  union { uint4 u; bf16x8 v; } c;
  c.u = *reinterpret_cast<const uint4*>(p);
  return c.v;
}
DEV bf16x8 lds16(const char* p) {
  union { uint4 u; bf16x8 v; } c;
  c.u = *reinterpret_cast<const uint4*>(p);
  return c.v;
}

// scatter one row-major 16B chunk (8 bf16) into a transposed [D][L_PAD]
// XOR-swizzled image: element j of row r at inner position c*8+j goes to
// image[d = c*8+j][r].
DEV void transpose_scatter(char* img, int LPAD, int r, int c, uint4 raw) {
  union { uint4 u; short s[8]; } v;
  v.u = raw;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int d = c * 8 + j;
    *reinterpret_cast<short*>(img + d * (LPAD * 2) + ((r * 2) ^ SWZBR(d, LPAD * 2))) = v.s[j];
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// B1: dS (global scratch) + dQ
// ---------------------------------------------------------------------------
template <int D, int NTMAX>
__global__ __launch_bounds__(256) void attn_bwd_ds_dq_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ probs,
    const bf16* __restrict__ dropm,  // nullptr = no dropout
    const bf16* __restrict__ dout, bf16* __restrict__ ds_out,
    bf16* __restrict__ dq, int B, int H, int Lq, int Lk, float scale) {
  constexpr int KCH = D / 8;
  const int HD = H * D;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();
  const int LK_PAD = (Lk + 31) & ~31;
  const int NT = LK_PAD / 16;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* KT_lds = smem;                               // [D][LK_PAD] transposed
  char* dS_lds = smem + D * LK_PAD * 2 + wid * 16 * LK_PAD * 2;  // per-wave

  // ---- stage K transposed ------------------------------------------------
  {
    const int rows_per_pass = blockDim.x / KCH;
    const int r0 = tid / KCH;
    const int c = tid % KCH;
    const long kbase = ((long)b * Lk) * HD + (long)h * D;
    const int npass = (LK_PAD + rows_per_pass - 1) / rows_per_pass;
#pragma unroll
    for (int pi = 0; pi < 8; ++pi) {
      if (pi >= npass) break;
      const int r = r0 + pi * rows_per_pass;
      if (r >= LK_PAD) break;
      const long rr = min(r, Lk - 1);
      const uint4 raw = *reinterpret_cast<const uint4*>(k + kbase + rr * HD + c * 8);
      transpose_scatter(KT_lds, LK_PAD, r, c, raw);
    }
  }
  __syncthreads();

  const int nstripes = (Lq + 15) / 16;
  const int col0 = lane & 15;
  for (int s = wid; s < nstripes; s += blockDim.x / WAVE) {
    const int qrow0 = s * 16;

    // ---- dPt = dO @ V^T (both operands straight from global) ------------
    f32x4 acc[NTMAX];
#pragma unroll
    for (int nt = 0; nt < NTMAX; ++nt) acc[nt] = {0.f, 0.f, 0.f, 0.f};
    {
      // A fragments: dO rows (contiguous d)
      bf16x8 ado[D / 32];
      const int arow = min(qrow0 + (lane & 15), Lq - 1);
      const long dobase = ((long)b * Lq + arow) * HD + (long)h * D + (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) ado[kk] = ld16(dout + dobase + kk * 32);
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int key = min(nt * 16 + col0, Lk - 1);
        const long vbase = ((long)b * Lk + key) * HD + (long)h * D + (lane >> 4) * 8;
#pragma unroll
        for (int kk = 0; kk < D / 32; ++kk) {
          const bf16x8 bv = ld16(v + vbase + kk * 32);
          acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[kk], bv, acc[nt], 0, 0, 0);
        }
      }
    }

    // ---- softmax backward on registers ----------------------------------
    // preload P (and dm) batched at clamped addresses (no guarded loads)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qrow0 + (lane >> 4) * 4 + r;
      const long pb = ((long)bh * Lq + min(row, Lq - 1)) * Lk;
      float pv[NTMAX], dmv[NTMAX];
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt)
        pv[nt] = bf2f(probs[pb + min(nt * 16 + col0, Lk - 1)]);
      if (dropm != nullptr) {
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
          dmv[nt] = bf2f(dropm[pb + min(nt * 16 + col0, Lk - 1)]);
      } else {
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) dmv[nt] = 1.f;
      }
      float dot = 0.f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int col = nt * 16 + col0;
        float dp = acc[nt][r] * dmv[nt];  // dP = dm .* dPt
        if (col >= Lk) { dp = 0.f; pv[nt] = 0.f; }
        acc[nt][r] = dp;
        dot += dp * pv[nt];
      }
      dot = group16_sum(dot);
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        acc[nt][r] = scale * pv[nt] * (acc[nt][r] - dot);  // dS
      }
      // dS -> per-wave row image (for dQ) + global scratch (for B2)
      const int prow = (lane >> 4) * 4 + r;
      char* rowb = dS_lds + prow * (LK_PAD * 2);
      const int psw = SWZBR(prow, LK_PAD * 2);
      const bool rowok = row < Lq;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int col = nt * 16 + col0;
        const unsigned short us = f2us(acc[nt][r]);
        *reinterpret_cast<short*>(rowb + ((col * 2) ^ psw)) = (short)us;
        if (rowok && col < Lk)
          ds_out[((long)bh * Lq + row) * Lk + col] = f2bf(acc[nt][r]);
      }
    }

    // ---- dQ = dS @ K (A: per-wave dS rows; B: K transposed image) --------
    f32x4 accq[D / 16];
#pragma unroll
    for (int nt = 0; nt < D / 16; ++nt) accq[nt] = {0.f, 0.f, 0.f, 0.f};
    const int parow = lane & 15;
    const char* pa = dS_lds + parow * (LK_PAD * 2);
    const int pasw = SWZBR(parow, LK_PAD * 2);
#pragma unroll
    for (int kk = 0; kk < NTMAX / 2; ++kk) {
      if (kk * 32 >= LK_PAD) break;
      const bf16x8 a = lds16(pa + (((kk * 64) + ((lane >> 4) * 16)) ^ pasw));
      const int keyoff = (kk * 64) + ((lane >> 4) * 16);
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) {
        const int d = nt * 16 + (lane & 15);
        const bf16x8 bk = lds16(KT_lds + d * (LK_PAD * 2) + (keyoff ^ SWZBR(d, LK_PAD * 2)));
        accq[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bk, accq[nt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qrow0 + (lane >> 4) * 4 + r;
      if (row < Lq) {
        const long qb = ((long)b * Lq + row) * HD + (long)h * D;
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt)
          dq[qb + nt * 16 + col0] = f2bf(accq[nt][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// B2: dK = dS^T @ Q, dV = P~^T @ dO
// LDS: QT [D][LQ_PAD] | dOT [D][LQ_PAD] | dST [LK_PAD][LQ_PAD] (shared) |
//      PT per-wave? no — PT [LK_PAD][LQ_PAD] shared too.
// ---------------------------------------------------------------------------
template <int D, int NQMAX>
__global__ __launch_bounds__(256) void attn_bwd_dk_dv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ dout,
    const bf16* __restrict__ probs, const bf16* __restrict__ dropm,
    const bf16* __restrict__ ds, bf16* __restrict__ dk,
    bf16* __restrict__ dv, int B, int H, int Lq, int Lk) {
  constexpr int KCH = D / 8;
  const int HD = H * D;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();
  const int LQ_PAD = (Lq + 31) & ~31;
  const int LK_PAD = (Lk + 31) & ~31;
  const int NQ = LQ_PAD / 16;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* QT_lds = smem;                       // [D][LQ_PAD]
  char* dOT_lds = QT_lds + D * LQ_PAD * 2;   // [D][LQ_PAD]
  char* dST_lds = dOT_lds + D * LQ_PAD * 2;  // [LK_PAD][LQ_PAD]
  char* PT_lds = dST_lds + LK_PAD * LQ_PAD * 2;

  // ---- stage Q and dO transposed (pad q-rows re-read the last row: their
  // columns multiply dS/P~ entries that are ZERO, see below) ---------------
  {
    const int rows_per_pass = blockDim.x / KCH;
    const int r0 = tid / KCH;
    const int c = tid % KCH;
    const long base = ((long)b * Lq) * HD + (long)h * D;
    const int npass = (LQ_PAD + rows_per_pass - 1) / rows_per_pass;
#pragma unroll
    for (int pi = 0; pi < 8; ++pi) {
      if (pi >= npass) break;
      const int r = r0 + pi * rows_per_pass;
      if (r >= LQ_PAD) break;
      const long rr = min(r, Lq - 1);
      transpose_scatter(QT_lds, LQ_PAD, r, c,
                        *reinterpret_cast<const uint4*>(q + base + rr * HD + c * 8));
      transpose_scatter(dOT_lds, LQ_PAD, r, c,
                        *reinterpret_cast<const uint4*>(dout + base + rr * HD + c * 8));
    }
  }
  // ---- stage dS^T and P~^T ([LK_PAD][LQ_PAD] row-major = key-major) ------
  // read 8-wide chunks of the [Lq,Lk] row-major scratch and scatter; pad
  // q-rows store EXPLICIT ZEROS (a clamped re-read would double-count).
  {
    // thread -> (qrow, key-chunk): Lk chunks of 8 along keys
    const int kch = LK_PAD / 8;
    const int rows_per_pass = blockDim.x / kch;
    const int r0 = tid / kch;
    const int c = tid % kch;
    for (int r = r0; r < LQ_PAD; r += rows_per_pass) {
      const bool rowok = r < Lq;
      const long pb = ((long)bh * Lq + min(r, Lq - 1)) * Lk;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int key = c * 8 + j;
        const long idx = pb + min(key, Lk - 1);
        const bool ok = rowok && key < Lk;
        const float dsv = ok ? bf2f(ds[idx]) : 0.f;
        float ptv = ok ? bf2f(probs[idx]) : 0.f;
        if (dropm != nullptr && ok) ptv *= bf2f(dropm[idx]);
        *reinterpret_cast<short*>(
            dST_lds + key * (LQ_PAD * 2) + ((r * 2) ^ SWZBR(key, LQ_PAD * 2))) = (short)f2us(dsv);
        *reinterpret_cast<short*>(
            PT_lds + key * (LQ_PAD * 2) + ((r * 2) ^ SWZBR(key, LQ_PAD * 2))) = (short)f2us(ptv);
      }
    }
  }
  __syncthreads();

  // ---- key stripes: dK rows and dV rows ----------------------------------
  const int nstripes = (Lk + 15) / 16;
  const int col0 = lane & 15;
  for (int s = wid; s < nstripes; s += blockDim.x / WAVE) {
    const int krow0 = s * 16;
    f32x4 acck[D / 16], accv[D / 16];
#pragma unroll
    for (int nt = 0; nt < D / 16; ++nt) {
      acck[nt] = {0.f, 0.f, 0.f, 0.f};
      accv[nt] = {0.f, 0.f, 0.f, 0.f};
    }
    const int arow = krow0 + (lane & 15);  // key row of the A fragment
    const char* dsbase = dST_lds + arow * (LQ_PAD * 2);
    const char* ptbase = PT_lds + arow * (LQ_PAD * 2);
    const int asw = SWZBR(arow, LQ_PAD * 2);
#pragma unroll
    for (int kk = 0; kk < NQMAX / 2; ++kk) {  // q-row slabs of 32
      if (kk * 32 >= LQ_PAD) break;
      const int qoff = (kk * 64) + ((lane >> 4) * 16);
      const bf16x8 a_ds = lds16(dsbase + (qoff ^ asw));
      const bf16x8 a_pt = lds16(ptbase + (qoff ^ asw));
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) {
        const int d = nt * 16 + (lane & 15);
        const bf16x8 bq = lds16(QT_lds + d * (LQ_PAD * 2) + (qoff ^ SWZBR(d, LQ_PAD * 2)));
        const bf16x8 bo = lds16(dOT_lds + d * (LQ_PAD * 2) + (qoff ^ SWZBR(d, LQ_PAD * 2)));
        acck[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_ds, bq, acck[nt], 0, 0, 0);
        accv[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_pt, bo, accv[nt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = krow0 + (lane >> 4) * 4 + r;
      if (row < Lk) {
        const long kb = ((long)b * Lk + row) * HD + (long)h * D;
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt) {
          dk[kb + nt * 16 + col0] = f2bf(acck[nt][r]);
          dv[kb + nt * 16 + col0] = f2bf(accv[nt][r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launcher: both kernels back-to-back on the stream
// ---------------------------------------------------------------------------
void launch_attention_bwd(const bf16* q, const bf16* k, const bf16* v,
                          const bf16* probs, const bf16* dropm,
                          const bf16* dout, bf16* ds_scratch, bf16* dq,
                          bf16* dk, bf16* dv, int B, int H, int Lq, int Lk,
                          int D, hipStream_t stream) {
  const float scale = 1.0f / sqrtf((float)D);
  const int LK_PAD = (Lk + 31) & ~31;
  const int LQ_PAD = (Lq + 31) & ~31;
  const dim3 grid(B * H);
  const size_t lds1 = (size_t)2 * (D * LK_PAD + 4 * 16 * LK_PAD);
  const size_t lds2 = (size_t)2 * (2 * D * LQ_PAD + 2 * LK_PAD * LQ_PAD);
#define LB1(DD, NTM)                                                          \
  hipLaunchKernelGGL((attn_bwd_ds_dq_kernel<DD, NTM>), grid, dim3(256), lds1, \
                     stream, q, k, v, probs, dropm, dout, ds_scratch, dq, B,  \
                     H, Lq, Lk, scale)
#define LB2(DD, NQM)                                                          \
  hipLaunchKernelGGL((attn_bwd_dk_dv_kernel<DD, NQM>), grid, dim3(256), lds2, \
                     stream, q, dout, probs, dropm, ds_scratch, dk, dv, B, H, \
                     Lq, Lk)
  if (D == 64) {
    if (LK_PAD <= 64) LB1(64, 4); else LB1(64, 8);
    if (LQ_PAD <= 64) LB2(64, 4); else LB2(64, 8);
  } else {
    if (LK_PAD <= 64) LB1(128, 4); else LB1(128, 8);
    if (LQ_PAD <= 64) LB2(128, 4); else LB2(128, 8);
  }
#undef LB1
#undef LB2
}
