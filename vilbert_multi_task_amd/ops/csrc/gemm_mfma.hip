// Hand-written CDNA4 MFMA GEMM for the ViLBERT hot projection/FFN shapes.
//
// Computes C = x @ W^T (+ bias) (+ residual) (+ GELU), bf16 in / f32
// accumulate / bf16 out — the torch nn.Linear contract with the epilogues
// the model needs fused (SURVEY.md §7.3's "GEMM-MFMA bf16 tile kernel";
// the reference's implicit cuBLAS GEMMs of worker.py:287-289).
//
// Why it exists even though hipBLASLt runs these shapes at ~1.07 PF/s
// (profiles/r06): (1) the north star requires the transformer GEMMs as
// hand-written MFMA kernels with an honest A/B; (2) the residual epilogue —
// hipBLASLt's beta=1 epilogue faults intermittently at some serving shapes
// ("write access to a read-only page", models/vilbert.py r1 note), while a
// hand-written epilogue trivially fuses y = x@W^T + b + res, removing a full
// HBM pass from the residual_ln that follows (12.1% of the serving step).
//
// Structure (guide §5 "canonical CDNA GEMM" + the 256² template parameters):
//   - tile 256(M) x 256(N) x BK=64, 512 threads = 8 waves as 2(M) x 4(N),
//     per-wave output 128x64 = acc[8][4] f32x4 fragments
//   - operands staged global->LDS with __builtin_amdgcn_global_load_lds
//     (16B/lane, lane-linear dest), double-buffered: 2 x (32KB A + 32KB B)
//     = 128 KiB LDS
//   - st_16x32 XOR swizzle (byte ^= ((byte>>9)&1)<<5) applied on the glds
//     SOURCE address and the ds_read_b128 offset (both-sides-or-neither,
//     guide §5.4 rule 21) — row-major [256][64] bf16 would otherwise put a
//     ds_read_b128 lane group 8-way on one bank slot
//   - both fragments read contiguous-K: A[m][k] from row-major x, and
//     B[k][n] = W[n][k] from row-major W (nn.Linear stores W as [N,K])
//   - ragged M/N handled by clamped loads + guarded stores; K % 64 == 0 and
//     N % 8 == 0 required (dispatch falls back to hipBLASLt otherwise)
//   - epilogue re-tiles C through LDS (the C fragment layout stores 2 bytes
//     per lane per instruction — 128 scalar stores/lane; via LDS it becomes
//     16 x b128 reads + 16 x 16B coalesced global stores), then adds the
//     residual streamed in b128 chunks. GELU is the exact erf form
//     (0.5*x*(1+erf(x/sqrt(2))), matching elementwise.hip:206).
//
// MFMA fragment maps: same as attention.hip (verified by mfma_probe):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j]
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C: lane l, reg r hold C[row = (l>>4)*4 + r][col = l&15]

#include "common.h"
#include <cstdlib>

#define KINVSQRT2 0.70710678118654752440f

// st_16x32 swizzle on a byte offset within a [256][64]-bf16 tile image
// (row stride 128 B): flips 32-byte half-lines when row bit 2 is set.
DEV int gswz(int o) { return o ^ (((o >> 9) & 1) << 5); }

// packed bf16 pair add (two lanes of a 32-bit word), f32 math, RNE repack
DEV unsigned int bfadd2(unsigned int a, unsigned int b) {
  const float lo = us2f((unsigned short)a) + us2f((unsigned short)b);
  const float hi = us2f((unsigned short)(a >> 16)) + us2f((unsigned short)(b >> 16));
  return (unsigned int)f2us(lo) | ((unsigned int)f2us(hi) << 16);
}

typedef __attribute__((address_space(3))) unsigned int* lds_u32p;

DEV bf16x8 lds_b128_g(const char* p) {
  union { uint4 u; bf16x8 v; } c;
  c.u = *reinterpret_cast<const uint4*>(p);
  return c.v;
}
typedef const __attribute__((address_space(1))) unsigned int* glob_u32p;

// one cooperative 8 KiB glds piece: 512 threads x 16 B, lane-linear LDS
DEV void glds16(const bf16* src_lane, char* lds_wave_base) {
  __builtin_amdgcn_global_load_lds(
      (glob_u32p)src_lane, (lds_u32p)lds_wave_base, 16, 0, 0);
}

DEV unsigned lds_byte_addr(const char* p) {
  return (unsigned)(size_t)(__attribute__((address_space(3))) const char*)p;
}

// inline-asm LDS-DMA (guide recipe): INVISIBLE to hipcc's waitcnt
// bookkeeping -- the compiler otherwise inserts s_waitcnt vmcnt(0) before
// every ds_read that might alias an outstanding glds, draining the 8-phase
// pipeline each phase. Completion is counted BY HAND with the schedule's
// vmcnt(4)/vmcnt(2) waits. M0 (LDS base) is saved/restored in-statement;
// per-lane dest = M0 + lane*16.
DEV void glds16_asm(const bf16* src_lane, unsigned lds_base) {
  unsigned keep;
  asm volatile(
      "s_mov_b32 %0, m0\n\t"
      "s_mov_b32 m0, %2\n\t"
      "s_nop 0\n\t"
      "global_load_lds_dwordx4 %1, off\n\t"
      "s_mov_b32 m0, %0"
      : "=&s"(keep)
      : "v"(src_lane), "s"(lds_base)
      : "memory");
}

template <bool GELU_, bool RES>
__global__ __launch_bounds__(512) void gemm256_kernel(
    const bf16* __restrict__ x,    // [M,K] row-major
    const bf16* __restrict__ w,    // [N,K] row-major (nn.Linear weight)
    const bf16* __restrict__ bias, // [N] or nullptr
    const bf16* __restrict__ res,  // [M,N] or nullptr (RES => non-null)
    bf16* __restrict__ out,        // [M,N]
    int M, int N, int K, int gx, int gy) {
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = __builtin_amdgcn_readfirstlane(wave_id());
  const int wm = wid >> 2;           // 0..1  (M half)
  const int wn = wid & 3;            // 0..3  (N quarter)

  // XCD-bijective block remap (guide §5: consecutive remapped ids share the
  // same W panel within one XCD's L2)
  int id = blockIdx.x;
  {
    const int nwg = gx * gy;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = id % 8;
    id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + id / 8;
  }
  const int blk_m = id % gx;
  const int blk_n = id / gx;
  const int row0 = blk_m * 256;  // x rows
  const int col0 = blk_n * 256;  // out cols = W rows

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [ Abuf0 32K | Abuf1 32K | Bbuf0 32K | Bbuf1 32K ] — computed by offset
  // (a pointer ARRAY of addrspace-cast bases fails to materialize)
#define ABUF(b) (smem + (b) * 32768)
#define BBUF(b) (smem + 65536 + (b) * 32768)

  // ---- per-thread staging geometry (constant across k-tiles) -------------
  // glds piece g covers image bytes [g*8192 + tid*16): the element that
  // belongs at lane-linear byte o is at swizzled position gswz(o)
  long srcA[4], srcB[4];  // element offsets (bytes) minus the k-tile advance
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    const int o = g * 8192 + tid * 16;
    const int os = gswz(o);
    const int r = os >> 7;         // row within the 256-row tile
    const int kb = os & 127;       // byte within the 64-element K slice
    srcA[g] = (long)min(row0 + r, M - 1) * (K * 2) + kb;
    srcB[g] = (long)min(col0 + r, N - 1) * (K * 2) + kb;
  }
  const char* xB = reinterpret_cast<const char*>(x);
  const char* wB = reinterpret_cast<const char*>(w);

  // ---- per-thread fragment read offsets (constant; buffer base varies) ---
  int offA[2][8], offB[2][4];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int kb = (kk * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
    for (int mt = 0; mt < 8; ++mt)
      offA[kk][mt] = gswz((wm * 128 + mt * 16 + (lane & 15)) * 128 + kb);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
      offB[kk][nt] = gswz((wn * 64 + nt * 16 + (lane & 15)) * 128 + kb);
  }

  // bias values for this lane's four 16-col tiles (preloaded, clamped)
  float bias_v[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    const int c = col0 + wn * 64 + nt * 16 + (lane & 15);
    bias_v[nt] = bias ? bf2f(bias[min(c, N - 1)]) : 0.f;
  }

  auto stage = [&](int buf, int t) {
    const long kadv = (long)t * 128;  // 64 bf16 per k-tile
    char* la = ABUF(buf) + wid * 1024;
    char* lb = BBUF(buf) + wid * 1024;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      glds16(reinterpret_cast<const bf16*>(xB + srcA[g] + kadv), la + g * 8192);
      glds16(reinterpret_cast<const bf16*>(wB + srcB[g] + kadv), lb + g * 8192);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mt = 0; mt < 8; ++mt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[mt][nt] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K >> 6;  // K % 64 == 0 enforced by the launcher

  // prologue: tile 0 into buf 0
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) stage(buf ^ 1, t + 1);  // flies under this tile's MFMA
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[8], bfr[4];
#pragma unroll
      for (int mt = 0; mt < 8; ++mt)
        af[mt] = lds_b128_g(ABUF(buf) + offA[kk][mt]);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        bfr[nt] = lds_b128_g(BBUF(buf) + offB[kk][nt]);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mt = 0; mt < 8; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mt], bfr[nt], acc[mt][nt], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // next tile's glds must have landed; this wave's LDS reads are done
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

  // ---- epilogue: bias (+GELU) on registers, re-tile through LDS, then ----
  // ---- (+residual) and coalesced 16 B stores -----------------------------
  // per-wave scratch = its 16 KiB slice of the (now free) staging LDS
  char* scratch = smem + wid * 16384;
#pragma unroll
  for (int mt = 0; mt < 8; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mt][nt][r] + bias_v[nt];
        if (GELU_) v = 0.5f * v * (1.0f + erff(v * KINVSQRT2));
        const int rl = mt * 16 + (lane >> 4) * 4 + r;  // row in wave tile
        const int cl = nt * 16 + (lane & 15);          // col in wave tile
        *reinterpret_cast<short*>(scratch + rl * 128 + cl * 2) =
            (short)f2us(v);
      }
    }
  }
  __syncthreads();

  const int mrow0 = row0 + wm * 128;
  const int ncol0 = col0 + wn * 64;
  // batch the residual loads first (unconditional, clamped), then combine —
  // a per-chunk load->wait->add chain serializes 16 HBM round trips, and a
  // short-array union here round-trips every chunk through scratch
  uint4 rv[16];
  if (RES) {
    const long last = (long)M * N - 8;
#pragma unroll
    for (int c = 0; c < 16; ++c) {
      const int o = c * 1024 + lane * 16;
      long gb = (long)min(mrow0 + (o >> 7), M - 1) * N + ncol0 + (o & 127) / 2;
      rv[c] = *reinterpret_cast<const uint4*>(res + (gb < last ? gb : last));
    }
  }
#pragma unroll
  for (int c = 0; c < 16; ++c) {
    const int o = c * 1024 + lane * 16;
    const int row = mrow0 + (o >> 7);
    const int col = ncol0 + (o & 127) / 2;
    if (row >= M) continue;
    uint4 v = *reinterpret_cast<uint4*>(scratch + o);
    const long gb = (long)row * N + col;
    if (col + 8 <= N) {
      if (RES) {
        v.x = bfadd2(v.x, rv[c].x);
        v.y = bfadd2(v.y, rv[c].y);
        v.z = bfadd2(v.z, rv[c].z);
        v.w = bfadd2(v.w, rv[c].w);
      }
      *reinterpret_cast<uint4*>(out + gb) = v;
    } else if (col < N) {  // ragged-N tail (edge blocks only)
      unsigned int arr[4] = {v.x, v.y, v.z, v.w};
      for (int j = 0; j < N - col && j < 8; ++j) {
        float vv = us2f((unsigned short)(arr[j >> 1] >> ((j & 1) * 16)));
        if (RES) vv += bf2f(res[gb + j]);
        out[gb + j] = f2bf(vv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 8-phase pipelined variant (guide §5 "256² 8-phase template" structure):
// each k-tile runs as 4 phases of {ds_read new fragments | stage 2 pieces of
// the NEXT k-tile | barrier | 16 MFMA | counted vmcnt | barrier}, so staging
// glds stay in flight ACROSS barriers (never drained to 0 in the loop — the
// counted wait is the whole gain per the guide: 8-phase-with-drain0 ≈
// 1-phase). Quadrant order (0,0),(0,1),(1,0),(1,1) over (mh,nh) halves of
// the wave's 128x64 output keeps one A-half + both B-halves live (64 frag
// VGPRs + 128 acc).
//
// Staging schedule (piece = 8 KiB = one cooperative glds = 64 rows of an
// operand image; ktile t stages ktile t+1 into buf^1):
//   ph0: B pieces 0,1   ph1: B pieces 2,3   ph2: A pieces 0,2   ph3: A 1,3
// Need times at ktile t+1: ph0 reads A piece 2*wm + B piece wn (covered by
// t.ph2/ph0-1); ph2 reads A pieces 1,3 (staged t.ph3).
// Waits (own-wave counters; the following barrier certifies block-wide):
//   end ph1: vmcnt(4)  -> A pieces 1,3 of THIS ktile landed (read at ph2)
//   end ph3: vmcnt(2)  -> B0-3 + A0,2 of the NEXT ktile landed (read ph0)
// ---------------------------------------------------------------------------
template <bool GELU_, bool RES, int SCHED = 0>
__global__ __launch_bounds__(512) void gemm256p_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const bf16* __restrict__ bias, const bf16* __restrict__ res,
    bf16* __restrict__ out, int M, int N, int K, int gx, int gy) {
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = __builtin_amdgcn_readfirstlane(wave_id());
  const int wm = wid >> 2;
  const int wn = wid & 3;

  int id = blockIdx.x;
  {
    const int nwg = gx * gy;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = id % 8;
    id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + id / 8;
  }
  const int blk_m = id % gx;
  const int blk_n = id / gx;
  const int row0 = blk_m * 256;
  const int col0 = blk_n * 256;

  extern __shared__ __attribute__((aligned(16))) char smem[];
#define ABUF(b) (smem + (b) * 32768)
#define BBUF(b) (smem + 65536 + (b) * 32768)

  long srcA[4], srcB[4];
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    const int o = g * 8192 + tid * 16;
    const int os = gswz(o);
    const int r = os >> 7;
    const int kb = os & 127;
    srcA[g] = (long)min(row0 + r, M - 1) * (K * 2) + kb;
    srcB[g] = (long)min(col0 + r, N - 1) * (K * 2) + kb;
  }
  const char* xB = reinterpret_cast<const char*>(x);
  const char* wB = reinterpret_cast<const char*>(w);

  int offA[2][8], offB[2][4];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int kb = (kk * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
    for (int mt = 0; mt < 8; ++mt)
      offA[kk][mt] = gswz((wm * 128 + mt * 16 + (lane & 15)) * 128 + kb);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
      offB[kk][nt] = gswz((wn * 64 + nt * 16 + (lane & 15)) * 128 + kb);
  }

  float bias_v[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    const int c = col0 + wn * 64 + nt * 16 + (lane & 15);
    bias_v[nt] = bias ? bf2f(bias[min(c, N - 1)]) : 0.f;
  }

  auto stageA = [&](int buf, int t, int p) {
    glds16_asm(reinterpret_cast<const bf16*>(xB + srcA[p] + (long)t * 128),
               lds_byte_addr(ABUF(buf) + p * 8192) + wid * 1024);
  };
  auto stageB = [&](int buf, int t, int p) {
    glds16_asm(reinterpret_cast<const bf16*>(wB + srcB[p] + (long)t * 128),
               lds_byte_addr(BBUF(buf) + p * 8192) + wid * 1024);
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mt = 0; mt < 8; ++mt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[mt][nt] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K >> 6;

  // prologue: whole ktile 0 into buf 0
#pragma unroll
  for (int p = 0; p < 4; ++p) { stageA(0, 0, p); stageB(0, 0, p); }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  bf16x8 a_[8], b0_[4], b1_[4];
  auto rdA = [&](int buf, int mh) {
#pragma unroll
    for (int mtl = 0; mtl < 4; ++mtl)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        a_[mtl * 2 + kk] = lds_b128_g(ABUF(buf) + offA[kk][mh * 4 + mtl]);
  };
  auto rdB = [&](int buf, int nh, bf16x8 (&dst)[4]) {
#pragma unroll
    for (int ntl = 0; ntl < 2; ++ntl)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        dst[ntl * 2 + kk] = lds_b128_g(BBUF(buf) + offB[kk][nh * 2 + ntl]);
  };
  auto mm = [&](int mh, int nh, bf16x8 (&bf_)[4]) {
    __builtin_amdgcn_s_setprio(1);
    // kk OUTER: 8 independent accumulators between reuses of one (mtl,ntl)
    // (kk-inner issued back-to-back dependent MFMAs on the same acc)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mtl = 0; mtl < 4; ++mtl)
#pragma unroll
        for (int ntl = 0; ntl < 2; ++ntl)
          acc[mh * 4 + mtl][nh * 2 + ntl] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_[mtl * 2 + kk], bf_[ntl * 2 + kk], acc[mh * 4 + mtl][nh * 2 + ntl], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };
  // whole-mh variant for the 2-phase-per-ktile schedule: 32 MFMAs per
  // cluster over both B halves (SCHED 3 uses 2 barriers per ktile, not 8)
  auto mm2 = [&](int mh) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mtl = 0; mtl < 4; ++mtl) {
#pragma unroll
        for (int ntl = 0; ntl < 2; ++ntl)
          acc[mh * 4 + mtl][ntl] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_[mtl * 2 + kk], b0_[ntl * 2 + kk], acc[mh * 4 + mtl][ntl], 0, 0, 0);
#pragma unroll
        for (int ntl = 0; ntl < 2; ++ntl)
          acc[mh * 4 + mtl][2 + ntl] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_[mtl * 2 + kk], b1_[ntl * 2 + kk], acc[mh * 4 + mtl][2 + ntl], 0, 0, 0);
      }
    __builtin_amdgcn_s_setprio(0);
  };
#define PH_BAR() __builtin_amdgcn_s_barrier()
#define LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")
  // pre-MFMA sync by schedule form (see launcher): SCHED 1 defers the lgkm
  // drain past the barrier (read latency spans it); SCHED 2 drops the
  // post-MFMA barrier (safe only with the drain BEFORE the remaining one:
  // a wave ahead in the next k-tile stages into the buffer a lagging
  // wave's un-drained reads still target)
#define PH_SYNC() do {                                                        \
    if (SCHED == 1) { PH_BAR(); LGKM0(); }                                    \
    else { LGKM0(); PH_BAR(); }                                               \
  } while (0)
#define PH_END() do { if (SCHED != 2) PH_BAR(); } while (0)

  int buf = 0;
  if (SCHED == 3) {
    for (int t = 0; t < ntiles; ++t) {
      const bool st = t + 1 < ntiles;
      const int nb = buf ^ 1;
      // ph0: mh=0 x both B halves (32 MFMAs)
      rdA(buf, 0); rdB(buf, 0, b0_); rdB(buf, 1, b1_);
      if (st) { stageB(nb, t + 1, 0); stageB(nb, t + 1, 1);
                stageB(nb, t + 1, 2); stageB(nb, t + 1, 3); }
      LGKM0(); PH_BAR();
      mm2(0);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // own A1,A3 landed
      PH_BAR();
      // ph1: mh=1 (A-half swap only)
      rdA(buf, 1);
      if (st) { stageA(nb, t + 1, 0); stageA(nb, t + 1, 2);
                stageA(nb, t + 1, 1); stageA(nb, t + 1, 3); }
      LGKM0(); PH_BAR();
      mm2(1);
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");  // next B0-3 + A0,A2
      PH_BAR();
      buf = nb;
    }
  } else
  for (int t = 0; t < ntiles; ++t) {
    const bool st = t + 1 < ntiles;
    const int nb = buf ^ 1;
    // ph0: (0,0)
    rdA(buf, 0); rdB(buf, 0, b0_);
    if (st) { stageB(nb, t + 1, 0); stageB(nb, t + 1, 1); }
    PH_SYNC();
    mm(0, 0, b0_);
    PH_END();
    // ph1: (0,1)
    rdB(buf, 1, b1_);
    if (st) { stageB(nb, t + 1, 2); stageB(nb, t + 1, 3); }
    PH_SYNC();
    mm(0, 1, b1_);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // own A1,A3 landed
    PH_END();
    // ph2: (1,0)
    rdA(buf, 1);
    if (st) { stageA(nb, t + 1, 0); stageA(nb, t + 1, 2); }
    PH_SYNC();
    mm(1, 0, b0_);
    PH_END();
    // ph3: (1,1)
    if (st) { stageA(nb, t + 1, 1); stageA(nb, t + 1, 3); }
    PH_SYNC();
    mm(1, 1, b1_);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");  // next ktile's B0-3,A0,A2
    PH_END();
    buf = nb;
  }
#undef PH_SYNC
#undef PH_END
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // drain tail stages
#undef PH_BAR
#undef LGKM0

  // ---- epilogue: identical to gemm256_kernel ------------------------------
  char* scratch = smem + wid * 16384;
  __builtin_amdgcn_s_barrier();
#pragma unroll
  for (int mt = 0; mt < 8; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mt][nt][r] + bias_v[nt];
        if (GELU_) v = 0.5f * v * (1.0f + erff(v * KINVSQRT2));
        const int rl = mt * 16 + (lane >> 4) * 4 + r;
        const int cl = nt * 16 + (lane & 15);
        *reinterpret_cast<short*>(scratch + rl * 128 + cl * 2) =
            (short)f2us(v);
      }
    }
  }
  __syncthreads();

  const int mrow0 = row0 + wm * 128;
  const int ncol0 = col0 + wn * 64;
  uint4 rv[16];
  if (RES) {
    const long last = (long)M * N - 8;
#pragma unroll
    for (int c = 0; c < 16; ++c) {
      const int o = c * 1024 + lane * 16;
      long gb = (long)min(mrow0 + (o >> 7), M - 1) * N + ncol0 + (o & 127) / 2;
      rv[c] = *reinterpret_cast<const uint4*>(res + (gb < last ? gb : last));
    }
  }
#pragma unroll
  for (int c = 0; c < 16; ++c) {
    const int o = c * 1024 + lane * 16;
    const int row = mrow0 + (o >> 7);
    const int col = ncol0 + (o & 127) / 2;
    if (row >= M) continue;
    uint4 v = *reinterpret_cast<uint4*>(scratch + o);
    const long gb = (long)row * N + col;
    if (col + 8 <= N) {
      if (RES) {
        v.x = bfadd2(v.x, rv[c].x);
        v.y = bfadd2(v.y, rv[c].y);
        v.z = bfadd2(v.z, rv[c].z);
        v.w = bfadd2(v.w, rv[c].w);
      }
      *reinterpret_cast<uint4*>(out + gb) = v;
    } else if (col < N) {
      unsigned int arr[4] = {v.x, v.y, v.z, v.w};
      for (int j = 0; j < N - col && j < 8; ++j) {
        float vv = us2f((unsigned short)(arr[j >> 1] >> ((j & 1) * 16)));
        if (RES) vv += bf2f(res[gb + j]);
        out[gb + j] = f2bf(vv);
      }
    }
  }
#undef ABUF
#undef BBUF
}

// ---------------------------------------------------------------------------
// 4-wave variant: the structure rocBLAS/Tensile's winning MT256x256x64 kernel
// uses on this chip (256 threads, giant register tile). Wave grid 2(M)x2(N),
// per-wave output 128x128 = acc[8][8] f32x4 = 256 accumulation registers —
// at __launch_bounds__(256,1) there is 1 wave/SIMD and the full 512-register
// unified file, so the accumulators live in AGPRs and the arch VGPRs carry
// fragments + addressing. vs the 8-wave kernel: 2x the MFMA work per LDS
// byte read (128 MFMAs vs 64 per wave per ktile against the same 16 KiB of
// A reads + half the B traffic), and ONE barrier per ktile instead of 8 —
// the r08 PMC profile showed gemm256p's waves parked 37% of cycles at the
// per-phase barriers.
// Schedule per ktile t: {16 glds -> buf^1 (tile t+1, in flight under this
// tile's 128 MFMAs) | 2x(16 ds_read + 64 MFMA) | vmcnt(0) | barrier}. The
// counted-wait subtlety of the 8-phase schedule is unnecessary here: at the
// single wait point the only outstanding vmem IS the prefetch, and it had
// the whole MFMA cluster (~2k cycles) to land.
// ---------------------------------------------------------------------------
template <bool GELU_, bool RES, bool ILV = false>
__global__ __launch_bounds__(256, 1) void gemm256w_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const bf16* __restrict__ bias, const bf16* __restrict__ res,
    bf16* __restrict__ out, int M, int N, int K, int gx, int gy) {
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = __builtin_amdgcn_readfirstlane(wave_id());
  const int wm = wid >> 1;  // 0..1 (M half)
  const int wn = wid & 1;   // 0..1 (N half)

  int id = blockIdx.x;
  {
    const int nwg = gx * gy;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = id % 8;
    id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + id / 8;
  }
  const int blk_m = id % gx;
  const int blk_n = id / gx;
  const int row0 = blk_m * 256;
  const int col0 = blk_n * 256;

  extern __shared__ __attribute__((aligned(16))) char smem[];
#define ABUF(b) (smem + (b) * 32768)
#define BBUF(b) (smem + 65536 + (b) * 32768)

  // staging geometry: piece g = 4 KiB (256 threads x 16 B), 8 pieces/operand
  long srcA[8], srcB[8];
#pragma unroll
  for (int g = 0; g < 8; ++g) {
    const int o = g * 4096 + tid * 16;
    const int os = gswz(o);
    const int r = os >> 7;
    const int kb = os & 127;
    srcA[g] = (long)min(row0 + r, M - 1) * (K * 2) + kb;
    srcB[g] = (long)min(col0 + r, N - 1) * (K * 2) + kb;
  }
  const char* xB = reinterpret_cast<const char*>(x);
  const char* wB = reinterpret_cast<const char*>(w);

  int offA[2][8], offB[2][8];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int kb = (kk * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
    for (int mt = 0; mt < 8; ++mt)
      offA[kk][mt] = gswz((wm * 128 + mt * 16 + (lane & 15)) * 128 + kb);
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
      offB[kk][nt] = gswz((wn * 128 + nt * 16 + (lane & 15)) * 128 + kb);
  }

  float bias_v[8];
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
    const int c = col0 + wn * 128 + nt * 16 + (lane & 15);
    bias_v[nt] = bias ? bf2f(bias[min(c, N - 1)]) : 0.f;
  }

  auto stage = [&](int buf, int t) {
    const long kadv = (long)t * 128;
#pragma unroll
    for (int g = 0; g < 8; ++g) {
      glds16_asm(reinterpret_cast<const bf16*>(xB + srcA[g] + kadv),
                 lds_byte_addr(ABUF(buf) + g * 4096) + wid * 1024);
      glds16_asm(reinterpret_cast<const bf16*>(wB + srcB[g] + kadv),
                 lds_byte_addr(BBUF(buf) + g * 4096) + wid * 1024);
    }
  };

  f32x4 acc[8][8];
#pragma unroll
  for (int mt = 0; mt < 8; ++mt)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) acc[mt][nt] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K >> 6;

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // VILBERT_GEMM_W_ILV=1: interleave the 16 staging glds 2-per-8-MFMAs
  // through the kk=0 cluster (sched_group_barrier pins) instead of issuing
  // them as one burst — evens out the per-ktile 64 KB HBM demand spike all
  // 256 CUs otherwise emit in lockstep at cluster start.
  auto cluster = [&](int cbuf, int t1, bool stg, bool spread) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[8], bfr[8];
#pragma unroll
      for (int mt = 0; mt < 8; ++mt)
        af[mt] = lds_b128_g(ABUF(cbuf) + offA[kk][mt]);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
        bfr[nt] = lds_b128_g(BBUF(cbuf) + offB[kk][nt]);
      if (stg && !spread && kk == 0) stage(cbuf ^ 1, t1);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mt = 0; mt < 8; ++mt) {
#pragma unroll
        for (int nt = 0; nt < 8; ++nt)
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mt], bfr[nt], acc[mt][nt], 0, 0, 0);
        if (stg && spread && kk == 0) {
          __builtin_amdgcn_sched_group_barrier(0x8 /*MFMA*/, 8, 0);
          const long kadv = (long)t1 * 128;
          glds16_asm(reinterpret_cast<const bf16*>(xB + srcA[mt] + kadv),
                     lds_byte_addr(ABUF(cbuf ^ 1) + mt * 4096) + wid * 1024);
          glds16_asm(reinterpret_cast<const bf16*>(wB + srcB[mt] + kadv),
                     lds_byte_addr(BBUF(cbuf ^ 1) + mt * 4096) + wid * 1024);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  };
  int buf = 0;
  for (int t = 0; t < ntiles - 1; ++t) {
    cluster(buf, t + 1, true, ILV);
    // prefetch landed (only outstanding vmem); all waves' reads of buf done
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }
  cluster(buf, 0, false, false);
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // ---- epilogue: per-wave 32 KiB scratch (128x128 bf16), re-tile + store --
  char* scratch = smem + wid * 32768;
#pragma unroll
  for (int mt = 0; mt < 8; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mt][nt][r] + bias_v[nt];
        if (GELU_) v = 0.5f * v * (1.0f + erff(v * KINVSQRT2));
        const int rl = mt * 16 + (lane >> 4) * 4 + r;
        const int cl = nt * 16 + (lane & 15);
        *reinterpret_cast<short*>(scratch + rl * 256 + cl * 2) =
            (short)f2us(v);
      }
    }
  }
  __syncthreads();

  const int mrow0 = row0 + wm * 128;
  const int ncol0 = col0 + wn * 128;
  // 32 chunks of 1 KiB; residual batched in halves of 16 to bound VGPRs
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    uint4 rv[16];
    if (RES) {
      const long last = (long)M * N - 8;
#pragma unroll
      for (int c = 0; c < 16; ++c) {
        const int o = (h * 16 + c) * 1024 + lane * 16;
        long gb = (long)min(mrow0 + (o >> 8), M - 1) * N + ncol0 + (o & 255) / 2;
        rv[c] = *reinterpret_cast<const uint4*>(res + (gb < last ? gb : last));
      }
    }
#pragma unroll
    for (int c = 0; c < 16; ++c) {
      const int o = (h * 16 + c) * 1024 + lane * 16;
      const int row = mrow0 + (o >> 8);
      const int col = ncol0 + (o & 255) / 2;
      if (row >= M) continue;
      uint4 v = *reinterpret_cast<uint4*>(scratch + o);
      const long gb = (long)row * N + col;
      if (col + 8 <= N) {
        if (RES) {
          v.x = bfadd2(v.x, rv[c].x);
          v.y = bfadd2(v.y, rv[c].y);
          v.z = bfadd2(v.z, rv[c].z);
          v.w = bfadd2(v.w, rv[c].w);
        }
        *reinterpret_cast<uint4*>(out + gb) = v;
      } else if (col < N) {
        unsigned int arr[4] = {v.x, v.y, v.z, v.w};
        for (int j = 0; j < N - col && j < 8; ++j) {
          float vv = us2f((unsigned short)(arr[j >> 1] >> ((j & 1) * 16)));
          if (RES) vv += bf2f(res[gb + j]);
          out[gb + j] = f2bf(vv);
        }
      }
    }
  }
#undef ABUF
#undef BBUF
}

void launch_gemm256(const bf16* x, const bf16* w, const bf16* bias,
                    const bf16* res, bf16* out, long M, long N, long K,
                    bool gelu, hipStream_t stream) {
  const int gx = (int)((M + 255) / 256);
  const int gy = (int)((N + 255) / 256);
  const dim3 grid(gx * gy);
  const size_t lds = 131072;
  // 8-phase pipelined schedule is the default; VILBERT_GEMM_PIPE=0 falls
  // back to the simple 2-phase loop (A/B + bisection lever)
  static const int pipe = [] {
    const char* e = getenv("VILBERT_GEMM_PIPE");
    return e ? atoi(e) : 1;  // 0 = 2-phase loop; 1/2/3 = 8-phase SCHED 0/1/2
  }();
#define L(KRN, G, R)                                                         \
  hipLaunchKernelGGL((KRN<G, R>), grid, dim3(512), lds, stream,              \
                     x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy)
#define LP(SCH)                                                              \
  do {                                                                       \
    if (gelu) { if (res) hipLaunchKernelGGL((gemm256p_kernel<true, true, SCH>), grid, dim3(512), lds, stream, x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy); \
                else hipLaunchKernelGGL((gemm256p_kernel<true, false, SCH>), grid, dim3(512), lds, stream, x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy); }  \
    else      { if (res) hipLaunchKernelGGL((gemm256p_kernel<false, true, SCH>), grid, dim3(512), lds, stream, x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy); \
                else hipLaunchKernelGGL((gemm256p_kernel<false, false, SCH>), grid, dim3(512), lds, stream, x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy); } \
  } while (0)
  if (pipe == 1) LP(0);
  else if (pipe == 2) LP(1);
  else if (pipe == 3) LP(2);
  else if (pipe == 4) LP(3);
  else if (pipe == 5 || pipe == 6) {  // 4-wave / AGPR variant (256 threads);
                                       // 6 = glds interleaved 2-per-8-MFMAs
#define LW(G, R, I)                                                          \
    hipLaunchKernelGGL((gemm256w_kernel<G, R, I>), grid, dim3(256), lds,     \
                       stream, x, w, bias, res, out, (int)M, (int)N, (int)K, gx, gy)
    if (pipe == 6) {
      if (gelu) { if (res) LW(true, true, true); else LW(true, false, true); }
      else      { if (res) LW(false, true, true); else LW(false, false, true); }
    } else {
      if (gelu) { if (res) LW(true, true, false); else LW(true, false, false); }
      else      { if (res) LW(false, true, false); else LW(false, false, false); }
    }
#undef LW
  }
  else {
    if (gelu) { if (res) L(gemm256_kernel, true, true); else L(gemm256_kernel, true, false); }
    else      { if (res) L(gemm256_kernel, false, true); else L(gemm256_kernel, false, false); }
  }
#undef LP
#undef L
}
