// Fused multi-head attention for the ViLBERT serving shapes, gfx950 MFMA. v2
//
// Replaces the reference's unfused QK^T-matmul / masked-softmax / PV-matmul
// chain (implicit cuBLAS+CUDA kernels) with ONE kernel per
// forward: S = QK^T (mfma_f32_16x16x32_bf16, K staged in LDS), row softmax
// entirely in registers (the whole key axis fits: Lk <= 128 for 101 regions /
// 38 tokens — SURVEY.md §5: the scale axis is batch, not sequence), then
// O = P·V with V staged TRANSPOSED [D][LK_PAD] so B-fragment reads are
// contiguous ds_read_b128 (transpose writes are conflict-free: contiguous
// bytes per VT row).
//
// v2 over v1 (profiles/r01_serving_b256_kernels.md: v1 = 143.6 us/call, 19.5%
// of the serving forward): +8-element row pads -> XOR swizzle
// (byte ^= (row&7)<<4) on K, VT and P (guide §6 G4), shrinking LDS
// 87 KB -> 80 KB = 2 workgroups/CU.
// NOTE on ds_read_b64_tr_b16: measured semantics (tr16_probe op, MI355X):
// lane l elem j = mem[addr_of_lane(16*(l>>4) + 4j) + (l&3)*2] — per 16-lane
// group only lanes 0/4/8/12's addresses are honored (16 distinct values per
// read), so it CANNOT feed a 16x16x32 B-fragment (128 distinct values per
// group); the transposed-image staging below is the right structure here.
//
// Layout contract: q,k,v and out are the flattened projection outputs
// [B, L, H*D]; one workgroup (4 waves) owns one (b,h) pair; each wave owns
// 16-query-row stripes.
//
// MFMA fragment maps (gfx950 v_mfma_f32_16x16x32_bf16, verified on-device by
// the mfma_probe op + tests/test_gpu_ops.py):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C: lane l, reg r hold C[row = (l>>4)*4 + r][col = l&15]

#include "common.h"
#include <cstdlib>

// e4m3 epilogue support (fp8 serving mode): emit the context tensor's
// quantized copy during the O store (delayed scaling; read-guarded amax)
DEV void attn_atomic_max_f32(float* addr, float v) {
  if (v <= *reinterpret_cast<volatile float*>(addr)) return;
  atomicMax(reinterpret_cast<unsigned int*>(addr), __float_as_uint(v));
}

#define ATTN_MAX_L 128  // max Lq/Lk this kernel serves (serving shapes <=101+pad)
// LDS XOR swizzle: include the row's HIGH bits so sibling rows 8 apart land
// on different 16B slots. PMC on MI355X showed the VT transpose writes were
// a 16-way conflict (19.6M SQ_LDS_BANK_CONFLICT per dispatch): for a fixed
// element index j, the 16 staging threads write d = c*8+j — d&7 == j for
// all of them, so a (row&7)-only swizzle degenerates. (row>>3)&7 varies
// with c and spreads them; it also kills the residual 2-way read conflicts
// on K/P (16 consecutive rows now map to 16 distinct slots).
#define SWZ(row) (((((row) & 7) ^ (((row) >> 3) & 7))) << 4)
// row-length-masked variant for images whose row is LK_PAD*2 bytes (the V
// transposed image and the P buffer): at LK_PAD=32 the plain SWZ (up to
// 112) escapes the 64-byte row -> write collisions. LK_PAD >= 64 rows are
// unaffected (mask keeps all bits).
#define SWZR(row, rbytes) (SWZ(row) & ((rbytes) - 1))

typedef __attribute__((ext_vector_type(4))) short s4_vec;
typedef __attribute__((address_space(3))) s4_vec* lds_v4s;

DEV bf16x8 load_bf16x8(const bf16* p) {
  union { uint4 u; bf16x8 v; } c;
  c.u = *reinterpret_cast<const uint4*>(p);
  return c.v;
}

DEV bf16x8 lds_b128(const char* p) {
  union { uint4 u; bf16x8 v; } c;
  c.u = *reinterpret_cast<const uint4*>(p);
  return c.v;
}

DEV void lds_store_b128(char* p, uint4 v) { *reinterpret_cast<uint4*>(p) = v; }

// SWAP=true: compute S^T = K Q^T instead of S = Q K^T (same loads, swapped
// MFMA operands). Payoffs: (1) the softmax row lives at a FIXED lane
// (q = lane&15) so the reduction is 2 xor-shuffles (16,32) instead of the
// 4-step group16 butterfly, and l^-1 normalization folds into P in-lane;
// (2) P transposes to the PV A-fragment layout with 4 ds_bpermute per
// 32-key step (cvt-pack pairs of adjacent keys, then a fixed lane remap) —
// NO P_lds at all: the per-wave P buffer (8 KB at D=64) disappears, LDS
// drops to K+V and occupancy rises 6 -> 10 WGs/CU for text-self.
// Index algebra (verified): S^T C-layout puts S[q = lane&15]
// [key = nt*16 + (lane>>4)*4 + r]; the PV A-frag wants lane (c = lane>>4,
// q) to hold keys kk*32 + c*8 + 0..7, i.e. source lane
// ((c&1)*2 + (j>>1))*16 + q, packed dword j&1, tile 2*kk + (c>>1).
// Gated to the inference fast path (no probs export, no dropout, no fp8).
template <int D, bool KGLOBAL, int NTMAX, bool FP8OUT = false, int THREADS = 256,
          bool SWAP = false, int MINW = 1>
__global__ __launch_bounds__(THREADS, MINW) void attn_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ kg,
    const bf16* __restrict__ vg, const bf16* __restrict__ mask,
    bf16* __restrict__ out, int B, int H, int Lq, int Lk, int mask_mode,
    float scale, int qs, int ks, int vs, int nsplit,
    unsigned char* __restrict__ out8 = nullptr,
    const float* __restrict__ fp8_scales = nullptr,
    float* __restrict__ fp8_amaxes = nullptr, int fp8_site = 0,
    bf16* __restrict__ probs_out = nullptr,
    const bf16* __restrict__ dropm = nullptr) {
  const float fp8_inv = FP8OUT ? 1.0f / fp8_scales[fp8_site] : 0.f;
  float fp8_amax = 0.f;
  // qs/ks/vs: row strides (elems) of q/k/v — [B,L,H*D] views into a fused
  // QKV (or KV) projection pass without any copy (stride 3*HD / 2*HD).
  // nsplit: query-stripe splits per (b,h) — each workgroup stages K/V and
  // owns 4 stripes (one per wave), so Lq=101 runs as 2 workgroups instead
  // of serializing 7 stripes over 4 waves (K/V re-staged per split; 64 KB
  // re-read stays L2-hot).
  constexpr int KCH = D / 8;  // 16B chunks per row
  const int HD = H * D;
  const int bh = blockIdx.x / nsplit;
  const int split = blockIdx.x % nsplit;
  const int b = bh / H;
  const int h = bh % H;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();

  const int LK_PAD = (Lk + 31) & ~31;
  const int NT = LK_PAD / 16;  // <= NTMAX score tiles per stripe

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // KGLOBAL: K fragments read straight from global (L2-resident slice) —
  // no K_lds, LDS drops to V+P = 48 KB for D=128 -> 3 workgroups/CU.
  char* K_lds = smem;                          // [LK_PAD][D] bf16, XOR-swizzled
  char* V_lds = smem + (KGLOBAL ? 0 : LK_PAD * D * 2);  // VT [D][LK_PAD]
  char* P_lds = V_lds + LK_PAD * D * 2 + wid * 16 * LK_PAD * 2;  // per-wave

  // ---- stage K (swizzled row-major) and V (tr-readable subtiles) ---------
  {
    constexpr int RPP = THREADS / KCH;  // rows per pass: 16 (D=128), 32 (D=64)
    // max passes is a TEMPLATE constant (LK_PAD <= NTMAX*16): sizing the
    // staging arrays [8] regardless made the allocator hold dead uint4 slots
    // live across the batch-issue window — 48 extra VGPRs for the D=64/NT=4
    // text-self template, capping occupancy at 4 waves/SIMD where LDS allows 6
    constexpr int NPMAX = (NTMAX * 16 + RPP - 1) / RPP;  // <= 8
    const int rows_per_pass = RPP;
    const int r0 = tid / KCH;
    const int c = tid % KCH;
    const long kbase0 = (long)b * Lk * ks + (long)h * D;
    const long vbase0 = (long)b * Lk * vs + (long)h * D;
    // issue EVERY staging load first (up to NPMAX passes x K+V), then the LDS
    // store pass: one batch of HBM round trips instead of one per pass
    const int npass = (LK_PAD + rows_per_pass - 1) / rows_per_pass;
    uint4 kr[NPMAX], vr[NPMAX];
    // unconditional clamped loads (no per-element r<Lk branch — §5 trap
    // (c)); pad rows re-read row Lk-1: K pads are masked to -3e38 in the
    // softmax and V pads multiply by P == 0
#pragma unroll
    for (int pi = 0; pi < NPMAX; ++pi) {
      if (pi >= npass) break;
      const long r = min(r0 + pi * rows_per_pass, Lk - 1);
      if (!KGLOBAL)
        kr[pi] = *reinterpret_cast<const uint4*>(kg + kbase0 + r * ks + c * 8);
      vr[pi] = *reinterpret_cast<const uint4*>(vg + vbase0 + r * vs + c * 8);
    }
#pragma unroll
    for (int pi = 0; pi < NPMAX; ++pi) {
      if (pi >= npass) break;
      const int r = r0 + pi * rows_per_pass;
      if (r >= LK_PAD) break;
      const uint4 kraw = kr[pi];
      const uint4 vraw = vr[pi];
      if (!KGLOBAL)
        lds_store_b128(K_lds + r * (D * 2) + ((c * 16) ^ SWZ(r)), kraw);
      // V transposed image [D][LK_PAD] (XOR-swizzled rows): the PV MFMA
      // B-fragment wants per-lane contiguous keys at fixed d, so transpose
      // at staging. Writes are conflict-free: for fixed j, consecutive
      // threads (consecutive r) write contiguous bytes of one VT row.
      // (ds_read_b64_tr_b16 was measured unusable here: per 16-lane group
      // it honors only lanes 0/4/8/12's addresses — 16 distinct values per
      // read vs the 128 a 16x16x32 B-fragment needs; see tr16_probe.)
      {
        union { uint4 u; short s[8]; } vv;
        vv.u = vraw;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = c * 8 + j;
          *reinterpret_cast<short*>(
              V_lds + d * (LK_PAD * 2) + ((r * 2) ^ SWZR(d, LK_PAD * 2))) = vv.s[j];
        }
      }
    }
  }
  __syncthreads();

  // ---- per-wave stripes of 16 query rows ---------------------------------
  const int nstripes = (Lq + 15) / 16;
  for (int s = split * (blockDim.x / WAVE) + wid; s < nstripes;
       s += nsplit * (blockDim.x / WAVE)) {
    const int qrow0 = s * 16;
    // Q A-fragments straight from global
    bf16x8 aq[D / 32];
    {
      const int row = min(qrow0 + (lane & 15), Lq - 1);
      const long qoff = ((long)b * Lq + row) * qs + (long)h * D + (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) aq[kk] = load_bf16x8(q + qoff + kk * 32);
    }

    // ---- S = Q K^T --------------------------------------------------------
    f32x4 acc_s[NTMAX];
#pragma unroll
    for (int nt = 0; nt < NTMAX; ++nt) acc_s[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < NTMAX; ++nt) {
      if (nt >= NT) break;
      const int key = nt * 16 + (lane & 15);
      if (KGLOBAL) {
        const long kb = (long)b * Lk * ks + (long)h * D +
                        (long)min(key, Lk - 1) * ks + (lane >> 4) * 8;
#pragma unroll
        for (int kk = 0; kk < D / 32; ++kk) {
          const bf16x8 bk = load_bf16x8(kg + kb + kk * 32);
          acc_s[nt] = SWAP ? __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                 bk, (bf16x8)aq[kk], acc_s[nt], 0, 0, 0)
                           : __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                 (bf16x8)aq[kk], bk, acc_s[nt], 0, 0, 0);
        }
      } else {
        const char* kbase = K_lds + key * (D * 2);
        const int ksw = SWZ(key);
#pragma unroll
        for (int kk = 0; kk < D / 32; ++kk) {
          const bf16x8 bk = lds_b128(kbase + (((kk * 64) + ((lane >> 4) * 16)) ^ ksw));
          acc_s[nt] = SWAP ? __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                 bk, (bf16x8)aq[kk], acc_s[nt], 0, 0, 0)
                           : __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                 (bf16x8)aq[kk], bk, acc_s[nt], 0, 0, 0);
        }
      }
    }

    if (SWAP) {
      // S^T layout: this lane owns query q = lane&15, keys
      // nt*16 + (lane>>4)*4 + r across (nt, r).
      const int q16 = lane & 15;
      const int g4 = (lane >> 4) * 4;
      // mask + scale (key-indexed preloads, unconditional clamped)
      if (mask_mode == 1) {
        const long mb = (long)b * Lk;
        float mv[NTMAX][4];
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            mv[nt][r] = bf2f(mask[mb + min(nt * 16 + g4 + r, Lk - 1)]);
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int key = nt * 16 + g4 + r;
            acc_s[nt][r] =
                (key < Lk) ? acc_s[nt][r] * scale + mv[nt][r] : -3.0e38f;
          }
      } else if (mask_mode == 2) {
        const int qrow = min(qrow0 + q16, Lq - 1);
        const long mb = ((long)b * Lq + qrow) * Lk;
        float mv[NTMAX][4];
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            mv[nt][r] = bf2f(mask[mb + min(nt * 16 + g4 + r, Lk - 1)]);
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int key = nt * 16 + g4 + r;
            acc_s[nt][r] =
                (key < Lk) ? acc_s[nt][r] * scale + mv[nt][r] : -3.0e38f;
          }
      } else {
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int key = nt * 16 + g4 + r;
            acc_s[nt][r] = (key < Lk) ? acc_s[nt][r] * scale : -3.0e38f;
          }
      }
      // softmax over keys: in-lane (nt, r) then across the 4 lane groups
      float mx = -3.0e38f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
#pragma unroll
        for (int r = 0; r < 4; ++r) mx = fmaxf(mx, acc_s[nt][r]);
      }
      mx = fmaxf(mx, __shfl_xor(mx, 16));
      mx = fmaxf(mx, __shfl_xor(mx, 32));
      float sum = 0.f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float p = __expf(acc_s[nt][r] - mx);
          acc_s[nt][r] = p;
          sum += p;
        }
      }
      sum += __shfl_xor(sum, 16);
      sum += __shfl_xor(sum, 32);
      const float inv = 1.0f / sum;
      // normalize in-lane (q matches this lane) and pack key pairs to bf16
      int pk[NTMAX][2];
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        pk[nt][0] = (int)f2us(acc_s[nt][0] * inv) |
                    ((int)f2us(acc_s[nt][1] * inv) << 16);
        pk[nt][1] = (int)f2us(acc_s[nt][2] * inv) |
                    ((int)f2us(acc_s[nt][3] * inv) << 16);
      }
      // PV with register-transposed P: 4 bpermute gather the A-fragment
      const int c = lane >> 4;
      const int idx0 = (((c & 1) * 2) * 16 + q16) * 4;       // groups 2(c&1)
      const int idx1 = (((c & 1) * 2 + 1) * 16 + q16) * 4;   // .. and +1
      f32x4 acc_o[D / 16];
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NTMAX / 2; ++kk) {
        if (kk * 32 >= LK_PAD) break;
        // the bpermute VALUE operand is evaluated in the SOURCE lane, so the
        // tile index must be wave-uniform per instruction: gather both tiles
        // of this 32-key step and select by the target's half (c>>1)
        union { int d[4]; bf16x8 v; } ap;
        const int lo0 = __builtin_amdgcn_ds_bpermute(idx0, pk[2 * kk][0]);
        const int lo1 = __builtin_amdgcn_ds_bpermute(idx0, pk[2 * kk][1]);
        const int lo2 = __builtin_amdgcn_ds_bpermute(idx1, pk[2 * kk][0]);
        const int lo3 = __builtin_amdgcn_ds_bpermute(idx1, pk[2 * kk][1]);
        const int hi0 = __builtin_amdgcn_ds_bpermute(idx0, pk[2 * kk + 1][0]);
        const int hi1 = __builtin_amdgcn_ds_bpermute(idx0, pk[2 * kk + 1][1]);
        const int hi2 = __builtin_amdgcn_ds_bpermute(idx1, pk[2 * kk + 1][0]);
        const int hi3 = __builtin_amdgcn_ds_bpermute(idx1, pk[2 * kk + 1][1]);
        const bool upper = (c >> 1) != 0;
        ap.d[0] = upper ? hi0 : lo0;
        ap.d[1] = upper ? hi1 : lo1;
        ap.d[2] = upper ? hi2 : lo2;
        ap.d[3] = upper ? hi3 : lo3;
        const int keyoff = (kk * 64) + ((lane >> 4) * 16);
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt) {
          const int d = nt * 16 + (lane & 15);
          const bf16x8 bv =
              lds_b128(V_lds + d * (LK_PAD * 2) + (keyoff ^ SWZR(d, LK_PAD * 2)));
          acc_o[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap.v, bv,
                                                              acc_o[nt], 0, 0, 0);
        }
      }
      // O store: C-layout rows are queries again; P was pre-normalized
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qrow0 + (lane >> 4) * 4 + r;
        if (row < Lq) {
          const long obase = ((long)b * Lq + row) * HD + (long)h * D;
#pragma unroll
          for (int nt = 0; nt < D / 16; ++nt)
            out[obase + nt * 16 + (lane & 15)] = f2bf(acc_o[nt][r]);
        }
      }
      continue;
    }

    // ---- mask + softmax (rows live across the 16-lane group) -------------
    // Mask values are preloaded UNCONDITIONALLY at clamped addresses, then
    // folded into acc_s in one register pass. The previous per-element
    // guarded loads (`if (col < Lk) sv += bf2f(mask[...])`) compiled to 32
    // dependent load->vmcnt(0) round trips per stripe per wave (guide §5
    // trap (c)) — the dominant stall of the whole kernel (r2 .s audit).
    const int col0 = lane & 15;
    if (mask_mode == 1) {
      const long mb = (long)b * Lk;
      float mv1[NTMAX];
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt)
        mv1[nt] = bf2f(mask[mb + min(nt * 16 + col0, Lk - 1)]);
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        const int col = nt * 16 + col0;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale + mv1[nt] : -3.0e38f;
      }
    } else if (mask_mode == 2) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = min(qrow0 + (lane >> 4) * 4 + r, Lq - 1);
        const long mb = ((long)b * Lq + row) * Lk;
        float mv2[NTMAX];
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
          mv2[nt] = bf2f(mask[mb + min(nt * 16 + col0, Lk - 1)]);
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) {
          const int col = nt * 16 + col0;
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale + mv2[nt] : -3.0e38f;
        }
      }
    } else {
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        const int col = nt * 16 + col0;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale : -3.0e38f;
      }
    }
    float inv_l[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -3.0e38f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        mx = fmaxf(mx, acc_s[nt][r]);
      }
      const float mrow = group16_max(mx);
      float sum = 0.f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        // pad cols hold -3e38: exp underflows to exactly 0, no guard needed
        const float p = __expf(acc_s[nt][r] - mrow);
        acc_s[nt][r] = p;
        sum += p;
      }
      inv_l[r] = 1.0f / group16_sum(sum);
    }

    // ---- P -> LDS (bf16, swizzled row-major; A-frag readable) -------------
    // training (dropm != nullptr): PV consumes the DROPPED P~ = dm .* P
    // while the probs export below stays pre-dropout (what the backward
    // kernels need). dm values preload batched at clamped addresses.
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int prow = (lane >> 4) * 4 + r;
      char* prow_base = P_lds + prow * (LK_PAD * 2);
      const int psw = SWZR(prow, LK_PAD * 2);
      float dmv[NTMAX];
      if (dropm != nullptr) {
        const int rowc = min(qrow0 + (lane >> 4) * 4 + r, Lq - 1);
        const long db = ((long)bh * Lq + rowc) * Lk;
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
          dmv[nt] = bf2f(dropm[db + min(nt * 16 + col0, Lk - 1)]);
      } else {
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) dmv[nt] = 1.f;
      }
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int col = nt * 16 + col0;
        *reinterpret_cast<short*>(prow_base + ((col * 2) ^ psw)) =
            (short)f2us(acc_s[nt][r] * dmv[nt]);
      }
    }
    // per-wave P buffer: same wave writes then reads (compiler inserts the
    // lgkm waits through the address dependence); no cross-wave sharing.

    // ---- optional attention-prob export (output_all_attention_masks=True,
    // worker.py:288): normalized softmax rows to [B,H,Lq,Lk] bf16 ----------
    if (probs_out != nullptr) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qrow0 + (lane >> 4) * 4 + r;
        if (row < Lq) {
          const long pbase = ((long)bh * Lq + row) * Lk;
#pragma unroll
          for (int nt = 0; nt < NTMAX; ++nt) {
            if (nt >= NT) break;
            const int col = nt * 16 + col0;
            if (col < Lk)
              probs_out[pbase + col] = f2bf(acc_s[nt][r] * inv_l[r]);
          }
        }
      }
    }

    // ---- O = P V (V fragments via hardware transpose reads) ---------------
    f32x4 acc_o[D / 16];
#pragma unroll
    for (int nt = 0; nt < D / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};
    const int parow = lane & 15;
    const char* pa_base = P_lds + parow * (LK_PAD * 2);
    const int pasw = SWZR(parow, LK_PAD * 2);
#pragma unroll
    for (int kk = 0; kk < NTMAX / 2; ++kk) {  // LK_PAD/32 <= NTMAX/2
      if (kk * 32 >= LK_PAD) break;
      const bf16x8 ap =
          lds_b128(pa_base + (((kk * 64) + ((lane >> 4) * 16)) ^ pasw));
      const int keyoff = (kk * 64) + ((lane >> 4) * 16);  // byte offset of keys
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) {
        const int d = nt * 16 + (lane & 15);
        const bf16x8 bv =
            lds_b128(V_lds + d * (LK_PAD * 2) + (keyoff ^ SWZR(d, LK_PAD * 2)));
        acc_o[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc_o[nt], 0, 0, 0);
      }
    }

    // ---- normalize + store O ---------------------------------------------
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qrow0 + (lane >> 4) * 4 + r;
      if (row < Lq) {
        const long obase = ((long)b * Lq + row) * HD + (long)h * D;
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt) {
          const float o = acc_o[nt][r] * inv_l[r];
          out[obase + nt * 16 + col0] = f2bf(o);
          if (FP8OUT) {
            fp8_amax = fmaxf(fp8_amax, fabsf(o));
            const float c = fminf(fmaxf(o * fp8_inv, -448.f), 448.f);
            int w = __builtin_amdgcn_cvt_pk_fp8_f32(c, c, 0, false);
            out8[obase + nt * 16 + col0] = (unsigned char)(w & 0xff);
          }
        }
      }
    }
    // NO __syncthreads here: P_lds is per-wave and stripe counts differ
    // across waves (a block-wide barrier inside this loop would deadlock).
  }
  if (FP8OUT) {
    fp8_amax = wave_max(fp8_amax);
    if (lane_id() == 0 && fp8_amax > 0.f)
      attn_atomic_max_f32(&fp8_amaxes[fp8_site], fp8_amax);
  }
}

// ---------------------------------------------------------------------------
// Pipelined variant (VILBERT_ATTN_PIPE=1): hide the V staging latency behind
// the first stripe's S/softmax compute. PMC (profiles/r06) shows the default
// kernel at 13% MFMA-busy — staging-latency bound — because ALL K+V HBM
// loads must land (and store to LDS) before the first MFMA issues. Here:
//   1. issue K loads, then V loads; store ONLY K; barrier   (V still flying)
//   2. each wave runs its FIRST stripe's S = QK^T + softmax + P->LDS
//      (needs only K_lds) while the V loads complete in the background
//   3. all threads store V transposed from registers; barrier
//   4. first stripe's O = P·V, then the remaining stripes run the full
//      unpipelined body (V resident by then).
// The phase barriers sit OUTSIDE the per-stripe loops (every thread reaches
// them exactly once), so unequal stripe counts across waves cannot deadlock.
// ---------------------------------------------------------------------------
template <int D, int NTMAX, bool FP8OUT = false>
__global__ __launch_bounds__(256) void attn_kernel_pipe(
    const bf16* __restrict__ q, const bf16* __restrict__ kg,
    const bf16* __restrict__ vg, const bf16* __restrict__ mask,
    bf16* __restrict__ out, int B, int H, int Lq, int Lk, int mask_mode,
    float scale, int qs, int ks, int vs, int nsplit,
    unsigned char* __restrict__ out8 = nullptr,
    const float* __restrict__ fp8_scales = nullptr,
    float* __restrict__ fp8_amaxes = nullptr, int fp8_site = 0) {
  const float fp8_inv = FP8OUT ? 1.0f / fp8_scales[fp8_site] : 0.f;
  float fp8_amax = 0.f;
  constexpr int KCH = D / 8;
  const int HD = H * D;
  const int bh = blockIdx.x;  // nsplit unused in the pipe variant
  const int b = bh / H;
  const int h = bh % H;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();

  const int LK_PAD = (Lk + 31) & ~31;
  const int NT = LK_PAD / 16;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* K_lds = smem;
  char* V_lds = smem + LK_PAD * D * 2;
  char* P_lds = V_lds + LK_PAD * D * 2 + wid * 16 * LK_PAD * 2;

  // ---- phase 1: K loads first, V loads after, store K only --------------
  const int rows_per_pass = blockDim.x / KCH;
  const int r0 = tid / KCH;
  const int c = tid % KCH;
  const long kbase0 = (long)b * Lk * ks + (long)h * D;
  const long vbase0 = (long)b * Lk * vs + (long)h * D;
  const int npass = (LK_PAD + rows_per_pass - 1) / rows_per_pass;
  uint4 kr[8], vr[8];
#pragma unroll
  for (int pi = 0; pi < 8; ++pi) {
    if (pi >= npass) break;
    const int r = r0 + pi * rows_per_pass;
    kr[pi] = make_uint4(0, 0, 0, 0);
    if (r < Lk)
      kr[pi] = *reinterpret_cast<const uint4*>(kg + kbase0 + (long)r * ks + c * 8);
  }
#pragma unroll
  for (int pi = 0; pi < 8; ++pi) {
    if (pi >= npass) break;
    const int r = r0 + pi * rows_per_pass;
    vr[pi] = make_uint4(0, 0, 0, 0);
    if (r < Lk)
      vr[pi] = *reinterpret_cast<const uint4*>(vg + vbase0 + (long)r * vs + c * 8);
  }
#pragma unroll
  for (int pi = 0; pi < 8; ++pi) {
    if (pi >= npass) break;
    const int r = r0 + pi * rows_per_pass;
    if (r >= LK_PAD) break;
    lds_store_b128(K_lds + r * (D * 2) + ((c * 16) ^ SWZ(r)), kr[pi]);
  }
  __syncthreads();  // K_lds ready; vr[] still in flight / in regs

  const int nstripes = (Lq + 15) / 16;
  const int nw = blockDim.x / WAVE;
  const int col0 = lane & 15;

  float inv_l[4];
  // ---- S phase for one stripe: S = QK^T, mask+softmax, P -> P_lds -------
  auto s_phase = [&](int qrow0) {
    bf16x8 aq[D / 32];
    {
      const int row = min(qrow0 + (lane & 15), Lq - 1);
      const long qoff = ((long)b * Lq + row) * qs + (long)h * D + (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) aq[kk] = load_bf16x8(q + qoff + kk * 32);
    }
    f32x4 acc_s[NTMAX];
#pragma unroll
    for (int nt = 0; nt < NTMAX; ++nt) acc_s[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < NTMAX; ++nt) {
      if (nt >= NT) break;
      const int key = nt * 16 + (lane & 15);
      const char* kbase = K_lds + key * (D * 2);
      const int ksw = SWZ(key);
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) {
        const bf16x8 bk = lds_b128(kbase + (((kk * 64) + ((lane >> 4) * 16)) ^ ksw));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            (bf16x8)aq[kk], bk, acc_s[nt], 0, 0, 0);
      }
    }
    // unconditional clamped mask preload (see base kernel note: the guarded
    // in-loop loads serialized into per-element vmcnt(0) round trips)
    if (mask_mode == 1) {
      const long mb = (long)b * Lk;
      float mv1[NTMAX];
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt)
        mv1[nt] = bf2f(mask[mb + min(nt * 16 + col0, Lk - 1)]);
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        const int col = nt * 16 + col0;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale + mv1[nt] : -3.0e38f;
      }
    } else if (mask_mode == 2) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = min(qrow0 + (lane >> 4) * 4 + r, Lq - 1);
        const long mb = ((long)b * Lq + row) * Lk;
        float mv2[NTMAX];
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt)
          mv2[nt] = bf2f(mask[mb + min(nt * 16 + col0, Lk - 1)]);
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) {
          const int col = nt * 16 + col0;
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale + mv2[nt] : -3.0e38f;
        }
      }
    } else {
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        const int col = nt * 16 + col0;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale : -3.0e38f;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -3.0e38f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        mx = fmaxf(mx, acc_s[nt][r]);
      }
      const float mrow = group16_max(mx);
      float sum = 0.f;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const float p = __expf(acc_s[nt][r] - mrow);
        acc_s[nt][r] = p;
        sum += p;
      }
      inv_l[r] = 1.0f / group16_sum(sum);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int prow = (lane >> 4) * 4 + r;
      char* prow_base = P_lds + prow * (LK_PAD * 2);
      const int psw = SWZR(prow, LK_PAD * 2);
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int col = nt * 16 + col0;
        *reinterpret_cast<short*>(prow_base + ((col * 2) ^ psw)) =
            (short)f2us(acc_s[nt][r]);
      }
    }
  };

  // ---- PV phase for one stripe: O = P·V, normalize, store ---------------
  auto pv_phase = [&](int qrow0) {
    f32x4 acc_o[D / 16];
#pragma unroll
    for (int nt = 0; nt < D / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};
    const int parow = lane & 15;
    const char* pa_base = P_lds + parow * (LK_PAD * 2);
    const int pasw = SWZR(parow, LK_PAD * 2);
#pragma unroll
    for (int kk = 0; kk < NTMAX / 2; ++kk) {
      if (kk * 32 >= LK_PAD) break;
      const bf16x8 ap =
          lds_b128(pa_base + (((kk * 64) + ((lane >> 4) * 16)) ^ pasw));
      const int keyoff = (kk * 64) + ((lane >> 4) * 16);
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) {
        const int d = nt * 16 + (lane & 15);
        const bf16x8 bv =
            lds_b128(V_lds + d * (LK_PAD * 2) + (keyoff ^ SWZR(d, LK_PAD * 2)));
        acc_o[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc_o[nt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qrow0 + (lane >> 4) * 4 + r;
      if (row < Lq) {
        const long obase = ((long)b * Lq + row) * HD + (long)h * D;
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt) {
          const float o = acc_o[nt][r] * inv_l[r];
          out[obase + nt * 16 + col0] = f2bf(o);
          if (FP8OUT) {
            fp8_amax = fmaxf(fp8_amax, fabsf(o));
            const float cc = fminf(fmaxf(o * fp8_inv, -448.f), 448.f);
            int w = __builtin_amdgcn_cvt_pk_fp8_f32(cc, cc, 0, false);
            out8[obase + nt * 16 + col0] = (unsigned char)(w & 0xff);
          }
        }
      }
    }
  };

  // ---- phase 2: first stripe's S while V loads land ----------------------
  const bool has0 = wid < nstripes;
  if (has0) s_phase(wid * 16);

  // ---- phase 3: V transposed stores from regs, then block-wide barrier ---
#pragma unroll
  for (int pi = 0; pi < 8; ++pi) {
    if (pi >= npass) break;
    const int r = r0 + pi * rows_per_pass;
    if (r >= LK_PAD) break;
    union { uint4 u; short s[8]; } vv;
    vv.u = vr[pi];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = c * 8 + j;
      *reinterpret_cast<short*>(
          V_lds + d * (LK_PAD * 2) + ((r * 2) ^ SWZR(d, LK_PAD * 2))) = vv.s[j];
    }
  }
  __syncthreads();  // V_lds ready for every wave

  // ---- phase 4: first stripe's PV, then the remaining stripes ------------
  if (has0) pv_phase(wid * 16);
  for (int s = wid + nw; s < nstripes; s += nw) {
    s_phase(s * 16);
    pv_phase(s * 16);
  }
  if (FP8OUT) {
    fp8_amax = wave_max(fp8_amax);
    if (lane_id() == 0 && fp8_amax > 0.f)
      attn_atomic_max_f32(&fp8_amaxes[fp8_site], fp8_amax);
  }
}

// ---------------------------------------------------------------------------
// v3: bh-loop with cross-iteration K/V prefetch (round-2 kernel).
//
// Why: PMC (profiles/r06) shows the per-(b,h) kernel at 13% MFMA-busy —
// staging-latency bound, because every workgroup cold-starts: ALL K+V HBM
// loads must land and be stored to LDS before its first MFMA. At B=1024 the
// op moves ~850 MB (HBM roofline ~135 us at 6.3 TB/s) but measured 315 us
// (43% of roofline): the fix is hiding the staging latency, not more MFMA.
//
// Structure (guide T14 async-STAGE split, applied ACROSS bh iterations):
// each workgroup walks bh pairs grid-stride with DOUBLE-BUFFERED K/V LDS.
// Per iteration:
//   1. preload this bh's Q rows + mask values into registers (oldest loads,
//      so the compute waits on vmcnt leave the prefetch in flight)
//   2. issue the NEXT bh's K/V global loads into registers (in flight
//      through the whole compute phase — ~10k cycles of MFMA/softmax cover
//      the ~1k-cycle HBM latency and the staging bandwidth)
//   3. compute all stripes from LDS[cur]
//   4. barrier; write the prefetched K (swizzled) / V (transposed) tiles to
//      LDS[cur^1]; barrier.
// All staging loads are ordinary register loads (NOT global_load_lds): the
// guide's mixed-kind trap ("beside glds hipcc waits vmcnt(0) for any
// ordinary load's use") would drain the prefetch at the first Q use.
// ---------------------------------------------------------------------------
template <int D, int NTMAX, int THREADS = 256>
__global__ __launch_bounds__(THREADS) void attn_bhloop_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ kg,
    const bf16* __restrict__ vg, const bf16* __restrict__ mask,
    bf16* __restrict__ out, int BH, int H, int Lq, int Lk, int mask_mode,
    float scale, int qs, int ks, int vs,
    bf16* __restrict__ probs_out = nullptr) {
  constexpr int KCH = D / 8;      // 16B chunks per row
  constexpr int NW = THREADS / WAVE;
  constexpr int SMAX = 512 / THREADS;  // stripes per wave (Lq <= 16*NW*SMAX = 128)
  const int HD = H * D;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();

  const int LK_PAD = (Lk + 31) & ~31;
  const int NT = LK_PAD / 16;
  const int kvbytes = LK_PAD * D * 2;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [K0 | V0 | K1 | V1 | P(per wave)]
  char* P_base = smem + 4 * kvbytes;

  // staging thread roles (same decomposition as v2)
  const int rows_per_pass = THREADS / KCH;
  const int r0 = tid / KCH;
  const int c = tid % KCH;
  const int npass = (LK_PAD + rows_per_pass - 1) / rows_per_pass;  // <= 8

  uint4 kr[8], vr[8];
  // UNCONDITIONAL loads at clamped rows: a per-element `if (r < Lk)` guard
  // makes hipcc branch around each load and wait per element (guide §5 trap
  // (c)). Pad rows re-read row Lk-1: K pad rows never matter (their score
  // cols are forced to -3e38) and V pad rows multiply by P == 0 — real data
  // instead of zeros avoids 0*NaN from garbage reads without any branch.
  auto issue_loads = [&](int bh) {
    const int b = bh / H, h = bh % H;
    const long kbase0 = (long)b * Lk * ks + (long)h * D;
    const long vbase0 = (long)b * Lk * vs + (long)h * D;
#pragma unroll
    for (int pi = 0; pi < 8; ++pi) {
      if (pi >= npass) break;
      const long r = min(r0 + pi * rows_per_pass, Lk - 1);
      kr[pi] = *reinterpret_cast<const uint4*>(kg + kbase0 + r * ks + c * 8);
      vr[pi] = *reinterpret_cast<const uint4*>(vg + vbase0 + r * vs + c * 8);
    }
  };
  auto write_tiles = [&](int buf) {
    char* K_lds = smem + 2 * buf * kvbytes;
    char* V_lds = K_lds + kvbytes;
#pragma unroll
    for (int pi = 0; pi < 8; ++pi) {
      if (pi >= npass) break;
      const int r = r0 + pi * rows_per_pass;
      if (r >= LK_PAD) break;
      lds_store_b128(K_lds + r * (D * 2) + ((c * 16) ^ SWZ(r)), kr[pi]);
      union { uint4 u; short s[8]; } vv;
      vv.u = vr[pi];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d = c * 8 + j;
        *reinterpret_cast<short*>(
            V_lds + d * (LK_PAD * 2) + ((r * 2) ^ SWZR(d, LK_PAD * 2))) = vv.s[j];
      }
    }
  };

  // prologue: stage the first bh synchronously
  if (blockIdx.x >= (unsigned)BH) return;  // grid never exceeds BH (launcher)
  issue_loads(blockIdx.x);
  write_tiles(0);
  __syncthreads();

  const int nstripes = (Lq + 15) / 16;
  const int col0 = lane & 15;

  for (int bh = blockIdx.x, it = 0; bh < BH; bh += gridDim.x, ++it) {
    const int buf = it & 1;
    const char* K_lds = smem + 2 * buf * kvbytes;
    const char* V_lds = K_lds + kvbytes;
    char* P_lds = P_base + wid * 16 * LK_PAD * 2;
    const int b = bh / H;

    // ---- 1. preload mask for this bh (oldest loads; one register set
    // serves every stripe and row in mode 1 — v2 reloaded them 4x per
    // stripe inside the softmax). Unconditional clamped loads (trap (c)).
    float mv[NTMAX];
    if (mask_mode == 1) {
      const long mb = (long)b * Lk;
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt)
        mv[nt] = bf2f(mask[mb + min(nt * 16 + col0, Lk - 1)]);
    } else {
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) mv[nt] = 0.f;
    }
    // Q fragments for this wave's stripes, loaded BEFORE the prefetch so
    // the compute phase contains no global loads at all — a compute-phase
    // load's wait would drain the (older, in-order-retired) prefetch queue
    bf16x8 aq[SMAX][D / 32];
#pragma unroll
    for (int si = 0; si < SMAX; ++si) {
      const int s = wid + si * NW;
      const int row = min(min(s, nstripes - 1) * 16 + (lane & 15), Lq - 1);
      const long qoff = ((long)b * Lq + row) * qs + (long)(bh % H) * D + (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) aq[si][kk] = load_bf16x8(q + qoff + kk * 32);
    }

    // ---- 2. prefetch the NEXT bh's K/V (stays in flight through compute) -
    const int nbh = bh + gridDim.x;
    if (nbh < BH) issue_loads(nbh);

    // ---- 3. stripes from LDS[buf] ----------------------------------------
#pragma unroll
    for (int si = 0; si < SMAX; ++si) {
      const int s = wid + si * NW;
      if (s >= nstripes) break;
      const int qrow0 = s * 16;

      f32x4 acc_s[NTMAX];
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) acc_s[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        if (nt >= NT) break;
        const int key = nt * 16 + (lane & 15);
        const char* kbase = K_lds + key * (D * 2);
        const int ksw = SWZ(key);
#pragma unroll
        for (int kk = 0; kk < D / 32; ++kk) {
          const bf16x8 bk = lds_b128(kbase + (((kk * 64) + ((lane >> 4) * 16)) ^ ksw));
          acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              (bf16x8)aq[si][kk], bk, acc_s[nt], 0, 0, 0);
        }
      }

#pragma unroll
      for (int nt = 0; nt < NTMAX; ++nt) {
        const int col = nt * 16 + col0;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_s[nt][r] = (col < Lk) ? acc_s[nt][r] * scale + mv[nt] : -3.0e38f;
      }
      float inv_l[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float mx = -3.0e38f;
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) {
          if (nt >= NT) break;
          mx = fmaxf(mx, acc_s[nt][r]);
        }
        const float mrow = group16_max(mx);
        float sum = 0.f;
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) {
          if (nt >= NT) break;
          const float p = __expf(acc_s[nt][r] - mrow);
          acc_s[nt][r] = p;
          sum += p;
        }
        inv_l[r] = 1.0f / group16_sum(sum);
      }

#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int prow = (lane >> 4) * 4 + r;
        char* prow_base = P_lds + prow * (LK_PAD * 2);
        const int psw = SWZR(prow, LK_PAD * 2);
#pragma unroll
        for (int nt = 0; nt < NTMAX; ++nt) {
          if (nt >= NT) break;
          const int col = nt * 16 + col0;
          *reinterpret_cast<short*>(prow_base + ((col * 2) ^ psw)) =
              (short)f2us(acc_s[nt][r]);
        }
      }
      if (probs_out != nullptr) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = qrow0 + (lane >> 4) * 4 + r;
          if (row < Lq) {
            const long pbase = ((long)bh * Lq + row) * Lk;
#pragma unroll
            for (int nt = 0; nt < NTMAX; ++nt) {
              if (nt >= NT) break;
              const int col = nt * 16 + col0;
              if (col < Lk)
                probs_out[pbase + col] = f2bf(acc_s[nt][r] * inv_l[r]);
            }
          }
        }
      }

      f32x4 acc_o[D / 16];
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};
      const int parow = lane & 15;
      const char* pa_base = P_lds + parow * (LK_PAD * 2);
      const int pasw = SWZR(parow, LK_PAD * 2);
#pragma unroll
      for (int kk = 0; kk < NTMAX / 2; ++kk) {
        if (kk * 32 >= LK_PAD) break;
        const bf16x8 ap =
            lds_b128(pa_base + (((kk * 64) + ((lane >> 4) * 16)) ^ pasw));
        const int keyoff = (kk * 64) + ((lane >> 4) * 16);
#pragma unroll
        for (int nt = 0; nt < D / 16; ++nt) {
          const int d = nt * 16 + (lane & 15);
          const bf16x8 bv =
              lds_b128(V_lds + d * (LK_PAD * 2) + (keyoff ^ SWZR(d, LK_PAD * 2)));
          acc_o[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc_o[nt], 0, 0, 0);
        }
      }

#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qrow0 + (lane >> 4) * 4 + r;
        if (row < Lq) {
          const long obase = ((long)b * Lq + row) * HD + (long)(bh % H) * D;
#pragma unroll
          for (int nt = 0; nt < D / 16; ++nt)
            out[obase + nt * 16 + col0] = f2bf(acc_o[nt][r] * inv_l[r]);
        }
      }
    }

    // ---- 4. publish the prefetched tiles into the other buffer -----------
    __syncthreads();  // all waves done reading LDS[buf] and P
    if (nbh < BH) write_tiles(buf ^ 1);
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// mfma layout probe: one-wave 16x16x32 product for on-device layout checks
// ---------------------------------------------------------------------------
// probe: out[l] = ds_bpermute(idx[l]*4, src[l]) — verifies full-wave64
// bpermute semantics + index addressing on this chip (SWAP-path gather).
__global__ void bperm_probe_kernel(const int* __restrict__ idx,
                                   const int* __restrict__ src,
                                   int* __restrict__ outv) {
  const int l = threadIdx.x;
  const int v = src[l];
  const int i = idx[l] * 4;
  outv[l] = __builtin_amdgcn_ds_bpermute(i, v);
}

__global__ void mfma_probe_kernel(const bf16* __restrict__ a,
                                  const bf16* __restrict__ b,
                                  float* __restrict__ c) {
  const int l = threadIdx.x;
  bf16x8 af, bf_;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = reinterpret_cast<const short*>(a)[(l & 15) * 32 + (l >> 4) * 8 + j];
    bf_[j] = reinterpret_cast<const short*>(b)[((l >> 4) * 8 + j) * 16 + (l & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf_, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
}

// ---------------------------------------------------------------------------
// tr16 semantics probe: lds[i] = i for i in [0,1024); each lane does one
// ds_read_b64_tr_b16. mode 0: per-lane canonical base (l&15)*2+(l>>4)*128 B;
// mode 1: uniform base 0. Output [64][4] int16 of the values each lane got.
// ---------------------------------------------------------------------------
__global__ void tr16_probe_kernel(short* __restrict__ outv, int mode) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int l = threadIdx.x;
  if (l < 64) {
    int elem_off = (mode == 0) ? ((l & 15) + (l >> 4) * 64) : 0;
    s4_vec v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4s)&lds[elem_off]);
#pragma unroll
    for (int j = 0; j < 4; ++j) outv[l * 4 + j] = v[j];
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

void launch_attention_impl(const bf16* q, const bf16* k, const bf16* v,
                           const bf16* mask, bf16* out, int B, int H, int Lq,
                           int Lk, int D, int mask_mode, int qs, int ks,
                           int vs, unsigned char* out8,
                           const float* fp8_scales, float* fp8_amaxes,
                           int fp8_site, hipStream_t stream,
                           bf16* probs_out = nullptr,
                           const bf16* dropm = nullptr) {
  const float scale = 1.0f / sqrtf((float)D);
  const int LK_PAD = (Lk + 31) & ~31;
  static const int kglobal_env = [] {
    const char* e = getenv("VILBERT_ATTN_KGLOBAL");
    return e ? atoi(e) : 0;  // measured: staged K beats L2-global K
  }();
  static const int nsplit_env = [] {
    const char* e = getenv("VILBERT_ATTN_NSPLIT");
    return e ? atoi(e) : 0;  // 0 = auto
  }();
  // measured on MI355X (B=256): nsplit>1 duplicates staging and loses
  int nsplit = nsplit_env > 0 ? nsplit_env : 1;
  static const int waves_env = [] {
    const char* e = getenv("VILBERT_ATTN_WAVES");
    return e ? atoi(e) : 0;  // 0 = default (4); 8 = double-occupancy D=128
  }();
  static const int pipe_env = [] {
    const char* e = getenv("VILBERT_ATTN_PIPE");
    return e ? atoi(e) : 0;  // 1 = V-staging pipelined behind stripe-0 S
  }();
  // read dynamically (not static) so tests/A-B harnesses can flip it per
  // call; getenv cost is noise next to a kernel launch
  const char* bhloop_e = getenv("VILBERT_ATTN_BHLOOP");
  const int bhloop_env = bhloop_e ? atoi(bhloop_e) : -1;  // -1 auto, 0 off, 1 force
  const int nwaves = (waves_env == 8 && D == 128 && !kglobal_env) ? 8 : 4;

  // ---- v3 bh-loop prefetch dispatch (the round-2 default for large BH) ----
  const int BH = B * H;
  const bool bhloop_ok = mask_mode <= 1 && !kglobal_env && nsplit == 1 &&
                         !pipe_env && nwaves == 4 && Lq <= 128 &&
                         out8 == nullptr && dropm == nullptr;
  // measured (r2 A/B @B1024): v3 first cut 0.42-0.70x of v2 — default OFF
  // until the prefetch actually pays; VILBERT_ATTN_BHLOOP=1 forces it for
  // iteration. (v2's 2-6 co-resident WGs/CU already overlap staging with
  // compute; v3's single-WG double-buffer must beat that to ship.)
  const bool use_bhloop = bhloop_ok && bhloop_env == 1;
  if (use_bhloop) {
    // [K0|V0|K1|V1|P x 8 waves]
    const size_t lds2 =
        (size_t)4 * LK_PAD * D * 2 + (size_t)8 * 16 * LK_PAD * 2;
    // 512 threads (8 waves): halves the per-thread prefetch register
    // footprint (the 256-thread variant held 64 staging VGPRs across the
    // whole compute — the allocator pushed them to AGPRs, and the
    // accvgpr copies forced per-load waits that killed the pipeline)
    int nres = (int)(163840 / lds2);
    nres = nres < 1 ? 1 : (nres > 4 ? 4 : nres);  // 512-thread WGs: <=4/CU
    const dim3 grid2(BH < nres * 256 ? BH : nres * 256);
#define LAUNCH_BHLOOP(DD, NTM)                                               \
    hipLaunchKernelGGL((attn_bhloop_kernel<DD, NTM, 512>), grid2, dim3(512), \
                       lds2, stream, q, k, v, mask, out, BH, H, Lq, Lk,      \
                       mask_mode, scale, qs, ks, vs, probs_out)
    const bool small2 = LK_PAD <= 64;
    if (D == 64) { if (small2) LAUNCH_BHLOOP(64, 4); else LAUNCH_BHLOOP(64, 8); }
    else         { if (small2) LAUNCH_BHLOOP(128, 4); else LAUNCH_BHLOOP(128, 8); }
#undef LAUNCH_BHLOOP
    return;
  }
  // VILBERT_ATTN_OCC=5|6: register-capped D=64/NT=4 instantiation (min
  // waves per EU via launch_bounds) — probes whether text-self's 45%-of-
  // roofline gap is workgroup-concurrency-bound (104+16 regs cap it at
  // 4 waves/SIMD where its 24 KB LDS would allow 6).
  const char* occ_e = getenv("VILBERT_ATTN_OCC");
  const int occ_env = occ_e ? atoi(occ_e) : 0;
  if (occ_env >= 5 && D == 64 && LK_PAD <= 64 && !kglobal_env && nsplit == 1 &&
      !pipe_env && out8 == nullptr && probs_out == nullptr && dropm == nullptr) {
    const size_t lds_occ = sizeof(bf16) * (size_t)(2 * LK_PAD * 64 + 4 * 16 * LK_PAD);
    const dim3 grido(B * H);
    if (occ_env >= 6)
      hipLaunchKernelGGL((attn_kernel<64, false, 4, false, 256, false, 6>),
                         grido, dim3(256), lds_occ, stream, q, k, v, mask, out,
                         B, H, Lq, Lk, mask_mode, scale, qs, ks, vs, 1,
                         nullptr, nullptr, nullptr, 0, nullptr, nullptr);
    else
      hipLaunchKernelGGL((attn_kernel<64, false, 4, false, 256, false, 5>),
                         grido, dim3(256), lds_occ, stream, q, k, v, mask, out,
                         B, H, Lq, Lk, mask_mode, scale, qs, ks, vs, 1,
                         nullptr, nullptr, nullptr, 0, nullptr, nullptr);
    return;
  }
  // swapped-S^T fast path (no P_lds; see the SWAP template doc above).
  // VILBERT_ATTN_SWAP: 1 force-on, 0 force-off, unset = measured default.
  const char* swap_e = getenv("VILBERT_ATTN_SWAP");
  const int swap_env = swap_e ? atoi(swap_e) : -1;
  const bool swap_ok = probs_out == nullptr && dropm == nullptr &&
                       out8 == nullptr && !kglobal_env && nsplit == 1 &&
                       !pipe_env && nwaves == 4;
  const bool use_swap = swap_ok && (swap_env == 1);  // default off until A/B'd
  if (use_swap) {
    const size_t lds_swap = sizeof(bf16) * (size_t)(2 * LK_PAD * D);
    const dim3 grids(B * H);
#define LAUNCH_SWAP(DD, NTM)                                                  \
    hipLaunchKernelGGL((attn_kernel<DD, false, NTM, false, 256, true>),       \
                       grids, dim3(256), lds_swap, stream, q, k, v, mask,     \
                       out, B, H, Lq, Lk, mask_mode, scale, qs, ks, vs, 1,    \
                       nullptr, nullptr, nullptr, 0, nullptr, nullptr)
    const bool smalls = LK_PAD <= 64;
    if (D == 64) { if (smalls) LAUNCH_SWAP(64, 4); else LAUNCH_SWAP(64, 8); }
    else         { if (smalls) LAUNCH_SWAP(128, 4); else LAUNCH_SWAP(128, 8); }
#undef LAUNCH_SWAP
    return;
  }
  // (K if staged) + V + per-wave P (all bf16)
  const size_t lds = sizeof(bf16) *
      (size_t)((kglobal_env ? 1 : 2) * LK_PAD * D + nwaves * 16 * LK_PAD);
  const dim3 grid(B * H * nsplit);
#define LAUNCH_ATTN(DD, KG, NTM)                                              \
  do {                                                                        \
    if (pipe_env && !KG && nsplit == 1 && nwaves == 4) {                      \
      if (out8)                                                               \
        hipLaunchKernelGGL((attn_kernel_pipe<DD, NTM, true>), grid, dim3(256),\
                           lds, stream, q, k, v, mask, out, B, H, Lq, Lk,     \
                           mask_mode, scale, qs, ks, vs, nsplit, out8,        \
                           fp8_scales, fp8_amaxes, fp8_site);                 \
      else                                                                    \
        hipLaunchKernelGGL((attn_kernel_pipe<DD, NTM, false>), grid,          \
                           dim3(256), lds, stream, q, k, v, mask, out, B, H,  \
                           Lq, Lk, mask_mode, scale, qs, ks, vs, nsplit);     \
    } else if (nwaves == 8) {                                                        \
      if (out8)                                                               \
        hipLaunchKernelGGL((attn_kernel<DD, KG, NTM, true, 512>), grid,       \
                           dim3(512), lds, stream, q, k, v, mask, out, B, H,  \
                           Lq, Lk, mask_mode, scale, qs, ks, vs, nsplit,      \
                           out8, fp8_scales, fp8_amaxes, fp8_site);           \
      else                                                                    \
        hipLaunchKernelGGL((attn_kernel<DD, KG, NTM, false, 512>), grid,      \
                           dim3(512), lds, stream, q, k, v, mask, out, B, H,  \
                           Lq, Lk, mask_mode, scale, qs, ks, vs, nsplit);     \
    } else if (out8)                                                          \
      hipLaunchKernelGGL((attn_kernel<DD, KG, NTM, true>), grid, dim3(256),   \
                         lds, stream, q, k, v, mask, out, B, H, Lq, Lk,       \
                         mask_mode, scale, qs, ks, vs, nsplit, out8,          \
                         fp8_scales, fp8_amaxes, fp8_site);                   \
    else                                                                      \
      hipLaunchKernelGGL((attn_kernel<DD, KG, NTM, false>), grid, dim3(256),  \
                         lds, stream, q, k, v, mask, out, B, H, Lq, Lk,       \
                         mask_mode, scale, qs, ks, vs, nsplit, nullptr,       \
                         nullptr, nullptr, 0, probs_out, dropm);              \
  } while (0)
  const bool small = LK_PAD <= 64;  // NTMAX=4 halves the accumulator VGPRs
  if (D == 64) {
    if (kglobal_env) { if (small) LAUNCH_ATTN(64, true, 4); else LAUNCH_ATTN(64, true, 8); }
    else             { if (small) LAUNCH_ATTN(64, false, 4); else LAUNCH_ATTN(64, false, 8); }
  } else if (D == 128) {
    if (kglobal_env) { if (small) LAUNCH_ATTN(128, true, 4); else LAUNCH_ATTN(128, true, 8); }
    else             { if (small) LAUNCH_ATTN(128, false, 4); else LAUNCH_ATTN(128, false, 8); }
  }
#undef LAUNCH_ATTN
}

void launch_attention(const bf16* q, const bf16* k, const bf16* v,
                      const bf16* mask, bf16* out, int B, int H, int Lq, int Lk,
                      int D, int mask_mode, int qs, int ks, int vs,
                      hipStream_t stream) {
  launch_attention_impl(q, k, v, mask, out, B, H, Lq, Lk, D, mask_mode, qs, ks,
                        vs, nullptr, nullptr, nullptr, 0, stream);
}

void launch_attention_probs(const bf16* q, const bf16* k, const bf16* v,
                            const bf16* mask, bf16* out, bf16* probs, int B,
                            int H, int Lq, int Lk, int D, int mask_mode,
                            int qs, int ks, int vs, hipStream_t stream) {
  launch_attention_impl(q, k, v, mask, out, B, H, Lq, Lk, D, mask_mode, qs, ks,
                        vs, nullptr, nullptr, nullptr, 0, stream, probs);
}

void launch_attention_train_fwd(const bf16* q, const bf16* k, const bf16* v,
                                const bf16* mask, const bf16* dropm,
                                bf16* out, bf16* probs, int B, int H, int Lq,
                                int Lk, int D, int mask_mode, int qs, int ks,
                                int vs, hipStream_t stream) {
  launch_attention_impl(q, k, v, mask, out, B, H, Lq, Lk, D, mask_mode, qs, ks,
                        vs, nullptr, nullptr, nullptr, 0, stream, probs, dropm);
}

void launch_attention_fp8out(const bf16* q, const bf16* k, const bf16* v,
                             const bf16* mask, bf16* out, unsigned char* out8,
                             const float* scales, float* amaxes, int site,
                             int B, int H, int Lq, int Lk, int D, int mask_mode,
                             int qs, int ks, int vs, hipStream_t stream) {
  launch_attention_impl(q, k, v, mask, out, B, H, Lq, Lk, D, mask_mode, qs, ks,
                        vs, out8, scales, amaxes, site, stream);
}

void launch_mfma_probe(const bf16* a, const bf16* b, float* c, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, a, b, c);
}
void launch_bperm_probe(const int* idx, const int* src, int* outv,
                        hipStream_t stream) {
  hipLaunchKernelGGL(bperm_probe_kernel, dim3(1), dim3(64), 0, stream, idx,
                     src, outv);
}

void launch_tr16_probe(short* out, int mode, hipStream_t stream) {
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream, out, mode);
}
