// hipBLASLt GEMM with fused bias+GELU epilogue.
//
// The FFN intermediate (768->3072 / 1024->1024, SURVEY.md §2.2 FFN row) is
// GEMM -> bias -> GELU; unfused that costs a full extra HBM round trip of
// the intermediate tensor plus a kernel launch (bias_gelu was 6.7% of the
// serving step, profiles/r01_serving_b256_kernels.md). Here the epilogue
// runs inside the hipBLASLt kernel (HIPBLASLT_EPILOGUE_GELU_BIAS).
//
// Note: hipBLASLt's GELU is the tanh approximation; BERT's reference is the
// erf form. Max divergence ~3e-3 absolute — below bf16 resolution of these
// activations; the fp32 oracle tests compare with that tolerance.
//
// y[M,N] = gelu(x[M,K] @ w[N,K]^T + b[N]) computed as column-major
// C[N,M] = op(A=w, T)[N,K] x op(B=x, N)[K,M].

#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>
#include <hipblaslt/hipblaslt-ext.hpp>
#include <cstdlib>

#include <map>
#include <vector>
#include <mutex>
#include <stdexcept>
#include <tuple>

#define HIPBLASLT_CHECK(x)                                          \
  do {                                                              \
    hipblasStatus_t s_ = (x);                                       \
    if (s_ != HIPBLAS_STATUS_SUCCESS)                               \
      throw std::runtime_error("hipblaslt error " + std::to_string(s_) + \
                               " at " #x);                          \
  } while (0)

namespace {

struct Plan {
  hipblasLtMatmulDesc_t op{};
  hipblasLtMatrixLayout_t a{}, b{}, c{};
  hipblasLtMatmulAlgo_t algo{};
  bool has_algo = false;
  float beta = 0.0f;
  std::vector<hipblasLtMatmulHeuristicResult_t> candidates;
};

hipblasLtHandle_t handle_once() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t t;
    HIPBLASLT_CHECK(hipblasLtCreate(&t));
    return t;
  }();
  return h;
}

std::map<std::tuple<long, long, long, int>, Plan>& plan_cache() {
  static std::map<std::tuple<long, long, long, int>, Plan> m;
  return m;
}
std::mutex& plan_mu() {
  static std::mutex m;
  return m;
}

// kind 0: epilogue GELU_BIAS, beta=0.  kind 1: epilogue BIAS, beta=1 (the
// C operand carries the residual: D = x@W^T + bias + residual).
// kind 2: epilogue BIAS, beta=0 (plain autotuned linear).
// kind 3: fp8(e4m3) A/B -> bf16 D, epilogue BIAS, per-tensor scales.
// kind 4: fp8 A/B -> fp8 D, epilogue GELU_BIAS, D scaled by *d_scale with
//         AMAX_D written to the site's amax slot (full-fp8 FFN chain).
Plan& get_plan(long M, long N, long K, int kind, void* workspace, size_t ws_bytes) {
  auto key = std::make_tuple(M, N, K, kind);
  auto& cache = plan_cache();
  auto it = cache.find(key);
  if (it != cache.end()) return it->second;

  const bool fp8_in = kind >= 3;
  const hipDataType ab_t = fp8_in ? HIP_R_8F_E4M3 : HIP_R_16BF;
  const hipDataType d_t = (kind == 4) ? HIP_R_8F_E4M3 : HIP_R_16BF;
  Plan p;
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&p.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  hipblasOperation_t ta = HIPBLAS_OP_T, tb = HIPBLAS_OP_N;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, sizeof(ta)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, sizeof(tb)));
  hipblasLtEpilogue_t epi =
      (kind == 0 || kind == 4) ? HIPBLASLT_EPILOGUE_GELU_BIAS
                               : HIPBLASLT_EPILOGUE_BIAS;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  if (fp8_in) {
    hipDataType bias_t = HIP_R_16BF;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_t, sizeof(bias_t)));
  }
  // A = w [K,N] col-major view of row-major [N,K], opA = T -> [N,K]
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.a, ab_t, K, N, K));
  // B = x [K,M] col-major view of row-major [M,K]
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.b, ab_t, K, M, K));
  // C = y [N,M] col-major view of row-major [M,N]
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.c, d_t, N, M, N));

  hipblasLtMatmulPreference_t pref;
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws_bytes, sizeof(ws_bytes)));
  // 32 heuristic candidates: autotune() times them once at eager warmup
  // (pre-capture), so a wider search costs warmup-only milliseconds.
  hipblasLtMatmulHeuristicResult_t results[32];
  int found = 0;
  HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
      handle_once(), p.op, p.a, p.b, p.c, p.c, pref, 32, results, &found));
  hipblasLtMatmulPreferenceDestroy(pref);
  if (found > 0) {
    p.algo = results[0].algo;
    p.has_algo = true;
    p.beta = kind == 1 ? 1.0f : 0.0f;
    p.candidates.assign(results, results + found);
    // Full-catalogue autotune (default OFF; VILBERT_GEMM_TUNE_FULL=1 opts
    // in): sweep every library algo supported for this problem. Measured
    // +1.1% steady-state @B1024 over the 32-candidate heuristic shortlist,
    // BUT the catalogue contains algos that pass matmulIsAlgoSupported and
    // then fault at run time ("Memory access fault" / "write access to a
    // read-only page") at several serving shapes — the round-1 beta=1
    // faults and the round-2 kind-2 faults are this one class. The
    // heuristic shortlist has never faulted; +1.1% is not worth a
    // process-killing fault lottery. scripts/debug_hipblaslt_algos.py pins
    // individual catalogue algos for isolation.
    static const bool tune_full = [] {
      const char* e = std::getenv("VILBERT_GEMM_TUNE_FULL");
      return e && e[0] == '1';
    }();
    if (tune_full) {
      std::vector<hipblasLtMatmulHeuristicResult_t> all;
      if (hipblaslt_ext::getAllAlgos(
              handle_once(), hipblaslt_ext::GemmType::HIPBLASLT_GEMM,
              HIPBLAS_OP_T, HIPBLAS_OP_N, ab_t, ab_t, d_t, d_t,
              HIPBLAS_COMPUTE_32F, all) == HIPBLAS_STATUS_SUCCESS) {
        float alpha = 1.0f, beta = p.beta;
        for (auto& c : all) {
          size_t need = 0;
          if (hipblaslt_ext::matmulIsAlgoSupported(
                  handle_once(), p.op, &alpha, p.a, p.b, &beta, p.c, p.c,
                  c.algo, need) == HIPBLAS_STATUS_SUCCESS &&
              need <= ws_bytes) {
            p.candidates.push_back(c);
            if (p.candidates.size() >= 400) break;
          }
        }
      }
    }
  }
  auto r = cache.emplace(key, p);
  return r.first->second;
}

// Time each heuristic candidate once (3 reps) and keep the fastest. Runs at
// plan creation — i.e. during eager warmup BEFORE hipGraph capture, where
// stream synchronization is legal. Scratch in/out buffers come from the
// caller's tensors (the timing matmuls write the real y, which the caller
// recomputes right after with the winning algo).
static void autotune(Plan& p, const void* x, const void* w, const void* bias,
                     const void* cmat, void* y, void* workspace,
                     size_t ws_bytes, hipStream_t stream) {
  if (p.candidates.empty()) return;
  // debug hook (scripts/debug_hipblaslt_algos.py): pin one candidate by
  // heuristic position instead of timing — lets a subprocess-per-algo
  // harness isolate which algo index faults at a given shape.
  static const int pin_pos = [] {
    const char* e = std::getenv("VILBERT_GEMM_TUNE_POS");
    return e ? atoi(e) : -1;
  }();
  if (pin_pos >= 0) {
    p.algo = p.candidates[std::min<size_t>(pin_pos, p.candidates.size() - 1)].algo;
    p.candidates.clear();
    return;
  }
  if (p.candidates.size() <= 1) return;
  float alpha = 1.0f, beta = p.beta;
  hipEvent_t ev0, ev1;
  if (hipEventCreate(&ev0) != hipSuccess) return;
  if (hipEventCreate(&ev1) != hipSuccess) { hipEventDestroy(ev0); return; }
  float best = 1e30f;
  hipblasLtMatmulAlgo_t best_algo = p.algo;
  for (auto& cand : p.candidates) {
    // correctness probe first
    if (hipblasLtMatmul(handle_once(), p.op, &alpha, w, p.a, x, p.b, &beta,
                        cmat, p.c, y, p.c, &cand.algo, workspace, ws_bytes,
                        stream) != HIPBLAS_STATUS_SUCCESS)
      continue;
    hipEventRecord(ev0, stream);
    for (int r = 0; r < 3; ++r)
      hipblasLtMatmul(handle_once(), p.op, &alpha, w, p.a, x, p.b, &beta,
                      cmat, p.c, y, p.c, &cand.algo, workspace, ws_bytes, stream);
    hipEventRecord(ev1, stream);
    hipEventSynchronize(ev1);
    float ms = 1e30f;
    hipEventElapsedTime(&ms, ev0, ev1);
    if (ms < best) { best = ms; best_algo = cand.algo; }
  }
  p.algo = best_algo;
  p.candidates.clear();  // tuned once
  hipEventDestroy(ev0);
  hipEventDestroy(ev1);
}

}  // namespace

// Shared driver. kind 0: gelu(x@W^T+b); kind 1: x@W^T+b+residual.
static int run_epilogue_gemm(int kind, const void* x, const void* w,
                             const void* bias, const void* residual, void* y,
                             long M, long N, long K, void* workspace,
                             size_t ws_bytes, hipStream_t stream) {
  try {
    std::lock_guard<std::mutex> lock(plan_mu());
    Plan& p = get_plan(M, N, K, kind, workspace, ws_bytes);
    if (!p.has_algo) return 1;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias)));
    // beta=0 kinds still need a VALID C pointer (hipBLASLt rejects NULL):
    // kind 2 used to pass residual=nullptr here, which made every direct
    // ext.linear_bias call fail ("no algo") and silently fall back to
    // torch F.linear in the wrapper — r2 .s/A-B audit finding
    const void* cmat = (kind == 0 || kind == 2) ? y : residual;
    int capturing = 0;
    hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
    if (hipStreamIsCapturing(stream, &st) == hipSuccess &&
        st != hipStreamCaptureStatusNone)
      capturing = 1;
    if (!capturing) autotune(p, x, w, bias, cmat, y, workspace, ws_bytes, stream);
    float alpha = 1.0f, beta = p.beta;
    HIPBLASLT_CHECK(hipblasLtMatmul(
        handle_once(), p.op, &alpha, w, p.a, x, p.b, &beta, cmat, p.c, y, p.c,
        &p.algo, workspace, ws_bytes, stream));
    return 0;
  } catch (const std::exception&) {
    return 1;
  }
}

int hipblaslt_linear_gelu(const void* x, const void* w, const void* bias,
                          void* y, long M, long N, long K, void* workspace,
                          size_t ws_bytes, hipStream_t stream) {
  return run_epilogue_gemm(0, x, w, bias, nullptr, y, M, N, K, workspace,
                           ws_bytes, stream);
}

int hipblaslt_linear_bias_add(const void* x, const void* w, const void* bias,
                              const void* residual, void* y, long M, long N,
                              long K, void* workspace, size_t ws_bytes,
                              hipStream_t stream) {
  return run_epilogue_gemm(1, x, w, bias, residual, y, M, N, K, workspace,
                           ws_bytes, stream);
}

int hipblaslt_linear_bias(const void* x, const void* w, const void* bias,
                          void* y, long M, long N, long K, void* workspace,
                          size_t ws_bytes, hipStream_t stream) {
  return run_epilogue_gemm(2, x, w, bias, nullptr, y, M, N, K, workspace,
                           ws_bytes, stream);
}

// fp8 matmuls: per-tensor scale pointers (device); kind 4 also writes the
// e4m3 D with *d_inv_scale applied and AMAX(D) to *amax_out.
static int run_fp8_gemm(int kind, const void* x8, const void* w8,
                        const void* bias, const void* w_scale,
                        const void* x_scale, const void* d_inv_scale,
                        void* amax_out, void* y, long M, long N, long K,
                        void* workspace, size_t ws_bytes, hipStream_t stream) {
  try {
    std::lock_guard<std::mutex> lock(plan_mu());
    Plan& p = get_plan(M, N, K, kind, workspace, ws_bytes);
    if (!p.has_algo) return 1;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias)));
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_A_SCALE_POINTER, &w_scale, sizeof(w_scale)));
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_B_SCALE_POINTER, &x_scale, sizeof(x_scale)));
    if (kind == 4) {
      HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
          p.op, HIPBLASLT_MATMUL_DESC_D_SCALE_POINTER, &d_inv_scale,
          sizeof(d_inv_scale)));
      HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
          p.op, HIPBLASLT_MATMUL_DESC_AMAX_D_POINTER, &amax_out,
          sizeof(amax_out)));
    }
    int capturing = 0;
    hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
    if (hipStreamIsCapturing(stream, &st) == hipSuccess &&
        st != hipStreamCaptureStatusNone)
      capturing = 1;
    if (!capturing) autotune(p, x8, w8, bias, y, y, workspace, ws_bytes, stream);
    float alpha = 1.0f, beta = 0.0f;
    HIPBLASLT_CHECK(hipblasLtMatmul(
        handle_once(), p.op, &alpha, w8, p.a, x8, p.b, &beta, y, p.c, y, p.c,
        &p.algo, workspace, ws_bytes, stream));
    return 0;
  } catch (const std::exception&) {
    return 1;
  }
}

int hipblaslt_fp8_linear(const void* x8, const void* w8, const void* bias,
                         const void* w_scale, const void* x_scale, void* y,
                         long M, long N, long K, void* workspace,
                         size_t ws_bytes, hipStream_t stream) {
  return run_fp8_gemm(3, x8, w8, bias, w_scale, x_scale, nullptr, nullptr, y,
                      M, N, K, workspace, ws_bytes, stream);
}

int hipblaslt_fp8_linear_gelu_fp8out(const void* x8, const void* w8,
                                     const void* bias, const void* w_scale,
                                     const void* x_scale,
                                     const void* d_inv_scale, void* amax_out,
                                     void* y8, long M, long N, long K,
                                     void* workspace, size_t ws_bytes,
                                     hipStream_t stream) {
  return run_fp8_gemm(4, x8, w8, bias, w_scale, x_scale, d_inv_scale, amax_out,
                      y8, M, N, K, workspace, ws_bytes, stream);
}
