// Torch operator registration for the gfx950 kernels.
// Pure C++/HIP (no Python.h): ops register under torch.ops.vilbert_amd and
// the .so is loaded in-tree via torch.ops.load_library (ops/hip_ext.py).

#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <torch/library.h>

#include <map>
#include <mutex>

#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;

// launchers (defined in the .hip files)
template <typename T>
void launch_residual_ln(const T*, const T*, const T*, const T*, T*, long, int,
                        float, hipStream_t);
template <typename T>
void launch_bias_gelu(const T*, const T*, T*, long, int, hipStream_t);
template <typename T>
void launch_embedding_ln(const long*, const long*, const long*, const T*,
                         const T*, const T*, const T*, const T*, T*, long, int,
                         float, hipStream_t);
void launch_attention(const bf16*, const bf16*, const bf16*, const bf16*, bf16*,
                      int, int, int, int, int, int, int, int, int, hipStream_t);
void launch_attention_probs(const bf16*, const bf16*, const bf16*, const bf16*,
                            bf16*, bf16*, int, int, int, int, int, int, int,
                            int, int, hipStream_t);
void launch_attention_train_fwd(const bf16*, const bf16*, const bf16*,
                                const bf16*, const bf16*, bf16*, bf16*, int,
                                int, int, int, int, int, int, int, int,
                                hipStream_t);
void launch_attention_bwd(const bf16*, const bf16*, const bf16*, const bf16*,
                          const bf16*, const bf16*, bf16*, bf16*, bf16*,
                          bf16*, int, int, int, int, int, hipStream_t);
void launch_attention_fp8out(const bf16*, const bf16*, const bf16*, const bf16*,
                             bf16*, unsigned char*, const float*, float*, int,
                             int, int, int, int, int, int, int, int, int,
                             hipStream_t);
void launch_mfma_probe(const bf16*, const bf16*, float*, hipStream_t);
void launch_bperm_probe(const int*, const int*, int*, hipStream_t);
void launch_gemm256(const bf16*, const bf16*, const bf16*, const bf16*, bf16*,
                    long, long, long, bool, hipStream_t);
void launch_adamw(bf16*, const void*, bool, float*, float*, float*, long,
                  float, float, float, float, float, long, hipStream_t);
void launch_nms_multiclass(const float*, const float*, const long*, float*, int,
                           int, float, float, hipStream_t);
void launch_tr16_probe(short*, int, hipStream_t);
int hipblaslt_linear_gelu(const void*, const void*, const void*, void*, long,
                          long, long, void*, size_t, hipStream_t);
int hipblaslt_linear_bias_add(const void*, const void*, const void*,
                              const void*, void*, long, long, long, void*,
                              size_t, hipStream_t);
int hipblaslt_linear_bias(const void*, const void*, const void*, void*, long,
                          long, long, void*, size_t, hipStream_t);
void launch_residual_ln_fp8(const bf16*, const bf16*, const bf16*, const bf16*,
                            bf16*, long, int, float, unsigned char*,
                            const float*, float*, int, hipStream_t);
void launch_bias_gelu_fp8(const bf16*, const bf16*, bf16*, long, int,
                          unsigned char*, const float*, float*, int,
                          hipStream_t);
void launch_quantize_fp8(const bf16*, unsigned char*, const float*, float*, int,
                         long, hipStream_t);
void launch_update_fp8_scales(float*, float*, float*, int, hipStream_t);
int hipblaslt_fp8_linear(const void*, const void*, const void*, const void*,
                         const void*, void*, long, long, long, void*, size_t,
                         hipStream_t);
int hipblaslt_fp8_linear_gelu_fp8out(const void*, const void*, const void*,
                                     const void*, const void*, const void*,
                                     void*, void*, long, long, long, void*,
                                     size_t, hipStream_t);
template <typename T>
void launch_roi_align(const T*, const float*, T*, int, int, int, int, int, int,
                      int, float, int, hipStream_t);
void launch_ln_fwd_train(const bf16*, const bf16*, const bf16*, const bf16*,
                         bf16*, bf16*, float*, float*, long, int, float,
                         hipStream_t);
void launch_ln_bwd_input(const bf16*, const bf16*, const float*, const float*,
                         const bf16*, bf16*, long, int, hipStream_t);
void launch_ln_bwd_param(const bf16*, const bf16*, const float*, const float*,
                         float*, float*, long, int, int, int, hipStream_t);
void launch_bias_gelu_bwd(const bf16*, const bf16*, bf16*, float*, long, int,
                          int, int, hipStream_t);

namespace {

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

at::Tensor residual_layer_norm(const at::Tensor& x,
                               const c10::optional<at::Tensor>& res,
                               const at::Tensor& w, const at::Tensor& b,
                               double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "residual_layer_norm: x");
  const int dim = (int)x.size(-1);
  const long rows = x.numel() / dim;
  auto y = at::empty_like(x);
  const bool hr = res.has_value();
  if (hr) TORCH_CHECK(res->is_contiguous() && res->sizes() == x.sizes());
  if (x.scalar_type() == at::kBFloat16) {
    launch_residual_ln<bf16>(
        (const bf16*)x.data_ptr(), hr ? (const bf16*)res->data_ptr() : nullptr,
        (const bf16*)w.data_ptr(), (const bf16*)b.data_ptr(),
        (bf16*)y.data_ptr(), rows, dim, (float)eps, cur_stream());
  } else if (x.scalar_type() == at::kFloat) {
    launch_residual_ln<float>(
        x.data_ptr<float>(), hr ? res->data_ptr<float>() : nullptr,
        w.data_ptr<float>(), b.data_ptr<float>(),
        y.data_ptr<float>(), rows, dim, (float)eps, cur_stream());
  } else {
    TORCH_CHECK(false, "residual_layer_norm: dtype must be bf16/f32");
  }
  return y;
}

at::Tensor bias_gelu(const at::Tensor& x, const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "bias_gelu: x");
  const int dim = (int)x.size(-1);
  const long n = x.numel();
  auto y = at::empty_like(x);
  const bool hb = bias.has_value();
  if (x.scalar_type() == at::kBFloat16) {
    launch_bias_gelu<bf16>((const bf16*)x.data_ptr(),
                           hb ? (const bf16*)bias->data_ptr() : nullptr,
                           (bf16*)y.data_ptr(), n, dim, cur_stream());
  } else if (x.scalar_type() == at::kFloat) {
    launch_bias_gelu<float>(x.data_ptr<float>(),
                            hb ? bias->data_ptr<float>() : nullptr,
                            y.data_ptr<float>(), n, dim, cur_stream());
  } else {
    TORCH_CHECK(false, "bias_gelu: dtype must be bf16/f32");
  }
  return y;
}

// accepts strided views along dim 1 (fused QKV projections): requires
// innermost stride 1 and batch stride == L * row-stride (16B-aligned rows).
static int row_stride_of(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.stride(2) == 1, name, ": innermost dim must be dense");
  const long rs = t.stride(1);
  TORCH_CHECK(t.stride(0) == t.size(1) * rs, name, ": batch stride mismatch");
  TORCH_CHECK(rs % 8 == 0, name, ": row stride must be 16B-aligned");
  return (int)rs;
}

at::Tensor attention(const at::Tensor& q, const at::Tensor& k,
                     const at::Tensor& v, int64_t heads,
                     const c10::optional<at::Tensor>& mask) {
  TORCH_CHECK(q.is_cuda());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attention: bf16 only");
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3, "attention: [B,L,H*D]");
  const int B = (int)q.size(0), Lq = (int)q.size(1), HD = (int)q.size(2);
  const int Lk = (int)k.size(1);
  const int qs = row_stride_of(q, "q"), kss = row_stride_of(k, "k"),
            vss = row_stride_of(v, "v");
  const int H = (int)heads;
  const int D = HD / H;
  TORCH_CHECK(HD % H == 0 && (D == 64 || D == 128), "attention: head_dim must be 64/128");
  TORCH_CHECK(Lq <= 128 && Lk <= 128, "attention: serving kernel handles L<=128");
  int mask_mode = 0;
  const bf16* mptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    auto& m = *mask;
    TORCH_CHECK(m.scalar_type() == at::kBFloat16, "attention: mask must be bf16");
    TORCH_CHECK(m.is_contiguous());
    const long mn = m.numel();
    if (mn == (long)B * Lk) mask_mode = 1;
    else if (mn == (long)B * Lq * Lk) mask_mode = 2;
    else TORCH_CHECK(false, "attention: mask numel must be B*Lk or B*Lq*Lk");
    mptr = (const bf16*)m.data_ptr();
  }
  auto out = at::empty({q.size(0), q.size(1), q.size(2)}, q.options());
  launch_attention((const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                   (const bf16*)v.data_ptr(), mptr,
                   (bf16*)out.data_ptr(), B, H, Lq, Lk, D, mask_mode,
                   qs, kss, vss, cur_stream());
  return out;
}

at::Tensor mfma_linear(const at::Tensor& x, const at::Tensor& w,
                       const c10::optional<at::Tensor>& bias,
                       const c10::optional<at::Tensor>& residual,
                       bool gelu) {
  // hand-written 256x256x64 MFMA GEMM (gemm_mfma.hip): y = x @ w^T
  // (+bias) (+residual) (+GELU) — the hipBLASLt-free hot-path GEMM with the
  // beta=1 residual epilogue hipBLASLt faults on
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16, "mfma_linear: bf16 cuda only");
  TORCH_CHECK(w.scalar_type() == at::kBFloat16);
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const long K = xc.size(-1);
  const long M = xc.numel() / K;
  const long N = wc.size(0);
  TORCH_CHECK(wc.size(-1) == K, "mfma_linear: K mismatch");
  TORCH_CHECK(K % 64 == 0, "mfma_linear: K % 64 != 0");
  TORCH_CHECK(N % 8 == 0, "mfma_linear: N % 8 != 0");
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto out = at::empty(sizes, x.options());
  const bf16* bp = nullptr;
  const bf16* rp = nullptr;
  at::Tensor rc;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->is_contiguous() && bias->numel() == N);
    bp = (const bf16*)bias->data_ptr();
  }
  if (residual.has_value() && residual->defined()) {
    rc = residual->contiguous();
    TORCH_CHECK(rc.numel() == M * N, "mfma_linear: residual shape");
    rp = (const bf16*)rc.data_ptr();
  }
  launch_gemm256((const bf16*)xc.data_ptr(), (const bf16*)wc.data_ptr(), bp,
                 rp, (bf16*)out.data_ptr(), M, N, K, gelu, cur_stream());
  return out;
}

void adamw_step(at::Tensor param, at::Tensor grad, at::Tensor master,
                at::Tensor m, at::Tensor v, double lr, double b1, double b2,
                double eps, double wd, int64_t step) {
  TORCH_CHECK(param.is_cuda() && param.scalar_type() == at::kBFloat16);
  TORCH_CHECK(master.scalar_type() == at::kFloat &&
              m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  const long n = param.numel();
  TORCH_CHECK(grad.numel() == n && master.numel() == n && m.numel() == n &&
              v.numel() == n);
  TORCH_CHECK(param.is_contiguous() && grad.is_contiguous() &&
              master.is_contiguous() && m.is_contiguous() && v.is_contiguous());
  const bool gb = grad.scalar_type() == at::kBFloat16;
  TORCH_CHECK(gb || grad.scalar_type() == at::kFloat);
  launch_adamw((bf16*)param.data_ptr(), grad.data_ptr(), gb,
               master.data_ptr<float>(), m.data_ptr<float>(),
               v.data_ptr<float>(), n, (float)lr, (float)b1, (float)b2,
               (float)eps, (float)wd, step, cur_stream());
}

std::tuple<at::Tensor, at::Tensor> attention_probs(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    int64_t heads, const c10::optional<at::Tensor>& mask) {
  // attention-map export path (worker.py:288 output_all_attention_masks=True):
  // same kernel, plus normalized softmax rows written to [B,H,Lq,Lk] bf16
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3);
  const int B = (int)q.size(0), Lq = (int)q.size(1), HD = (int)q.size(2);
  const int Lk = (int)k.size(1);
  const int H = (int)heads;
  const int D = HD / H;
  TORCH_CHECK(HD % H == 0 && (D == 64 || D == 128));
  TORCH_CHECK(Lq <= 128 && Lk <= 128);
  const int qs = row_stride_of(q, "q"), kss = row_stride_of(k, "k"),
            vss = row_stride_of(v, "v");
  int mask_mode = 0;
  const bf16* mptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    TORCH_CHECK(mask->scalar_type() == at::kBFloat16 && mask->is_contiguous());
    const long mn = mask->numel();
    if (mn == (long)B * Lk) mask_mode = 1;
    else if (mn == (long)B * Lq * Lk) mask_mode = 2;
    else TORCH_CHECK(false, "attention_probs: bad mask shape");
    mptr = (const bf16*)mask->data_ptr();
  }
  auto out = at::empty({q.size(0), q.size(1), q.size(2)}, q.options());
  auto probs = at::empty({(long)B, (long)H, (long)Lq, (long)Lk}, q.options());
  launch_attention_probs(
      (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
      (const bf16*)v.data_ptr(), mptr, (bf16*)out.data_ptr(),
      (bf16*)probs.data_ptr(), B, H, Lq, Lk, D, mask_mode, qs, kss, vss,
      cur_stream());
  return {out, probs};
}

std::tuple<at::Tensor, at::Tensor> attention_train_fwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    int64_t heads, const c10::optional<at::Tensor>& mask,
    const c10::optional<at::Tensor>& dropm) {
  // training forward: context uses the DROPPED probs, the returned probs
  // tensor is pre-dropout (what attention_bwd consumes)
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  const int B = (int)q.size(0), Lq = (int)q.size(1), HD = (int)q.size(2);
  const int Lk = (int)k.size(1);
  const int H = (int)heads;
  const int D = HD / H;
  TORCH_CHECK(HD % H == 0 && (D == 64 || D == 128));
  TORCH_CHECK(Lq <= 128 && Lk <= 128);
  const int qs = row_stride_of(q, "q"), kss = row_stride_of(k, "k"),
            vss = row_stride_of(v, "v");
  int mask_mode = 0;
  const bf16* mptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    TORCH_CHECK(mask->scalar_type() == at::kBFloat16 && mask->is_contiguous());
    const long mn = mask->numel();
    if (mn == (long)B * Lk) mask_mode = 1;
    else if (mn == (long)B * Lq * Lk) mask_mode = 2;
    else TORCH_CHECK(false, "attention_train_fwd: bad mask shape");
    mptr = (const bf16*)mask->data_ptr();
  }
  const bf16* dmp = nullptr;
  if (dropm.has_value() && dropm->defined()) {
    TORCH_CHECK(dropm->is_contiguous() &&
                dropm->numel() == (long)B * H * Lq * Lk);
    dmp = (const bf16*)dropm->data_ptr();
  }
  auto out = at::empty({q.size(0), q.size(1), q.size(2)}, q.options());
  auto probs = at::empty({(long)B, (long)H, (long)Lq, (long)Lk}, q.options());
  launch_attention_train_fwd(
      (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
      (const bf16*)v.data_ptr(), mptr, dmp, (bf16*)out.data_ptr(),
      (bf16*)probs.data_ptr(), B, H, Lq, Lk, D, mask_mode, qs, kss, vss,
      cur_stream());
  return {out, probs};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> attention_bwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& probs, const c10::optional<at::Tensor>& dropm,
    const at::Tensor& dout, int64_t heads) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(probs.is_contiguous());
  const int B = (int)q.size(0), Lq = (int)q.size(1), HD = (int)q.size(2);
  const int Lk = (int)k.size(1);
  const int H = (int)heads;
  const int D = HD / H;
  TORCH_CHECK(HD % H == 0 && (D == 64 || D == 128));
  auto dc = dout.contiguous();
  const bf16* dmp = nullptr;
  if (dropm.has_value() && dropm->defined()) {
    TORCH_CHECK(dropm->is_contiguous());
    dmp = (const bf16*)dropm->data_ptr();
  }
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto ds = at::empty({(long)B, (long)H, (long)Lq, (long)Lk}, q.options());
  launch_attention_bwd(
      (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
      (const bf16*)v.data_ptr(), (const bf16*)probs.data_ptr(), dmp,
      (const bf16*)dc.data_ptr(), (bf16*)ds.data_ptr(),
      (bf16*)dq.data_ptr(), (bf16*)dk.data_ptr(), (bf16*)dv.data_ptr(), B, H,
      Lq, Lk, D, cur_stream());
  return {dq, dk, dv};
}

std::tuple<at::Tensor, at::Tensor> attention_fp8out(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    int64_t heads, const c10::optional<at::Tensor>& mask,
    const at::Tensor& scales, const at::Tensor& amaxes, int64_t site) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  const int B = (int)q.size(0), Lq = (int)q.size(1), HD = (int)q.size(2);
  const int Lk = (int)k.size(1);
  const int H = (int)heads;
  const int D = HD / H;
  TORCH_CHECK(HD % H == 0 && (D == 64 || D == 128));
  TORCH_CHECK(Lq <= 128 && Lk <= 128);
  const int qs = row_stride_of(q, "q"), kss = row_stride_of(k, "k"),
            vss = row_stride_of(v, "v");
  int mask_mode = 0;
  const bf16* mptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    const long mn = mask->numel();
    if (mn == (long)B * Lk) mask_mode = 1;
    else if (mn == (long)B * Lq * Lk) mask_mode = 2;
    else TORCH_CHECK(false, "attention_fp8out: bad mask shape");
    mptr = (const bf16*)mask->data_ptr();
  }
  auto out = at::empty({q.size(0), q.size(1), q.size(2)}, q.options());
  auto out8 = at::empty(out.sizes(), out.options().dtype(at::kFloat8_e4m3fn));
  launch_attention_fp8out(
      (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
      (const bf16*)v.data_ptr(), mptr, (bf16*)out.data_ptr(),
      (unsigned char*)out8.data_ptr(), scales.data_ptr<float>(),
      amaxes.data_ptr<float>(), (int)site, B, H, Lq, Lk, D, mask_mode, qs,
      kss, vss, cur_stream());
  return {out, out8};
}

at::Tensor embedding_ln(const at::Tensor& ids, const at::Tensor& pos_ids,
                        const at::Tensor& type_ids, const at::Tensor& word_w,
                        const at::Tensor& pos_w, const at::Tensor& type_w,
                        const at::Tensor& ln_w, const at::Tensor& ln_b,
                        double eps) {
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == at::kLong);
  const int dim = (int)word_w.size(1);
  const long tokens = ids.numel();
  auto y = at::empty({ids.size(0), ids.size(1), (long)dim}, word_w.options());
  if (word_w.scalar_type() == at::kBFloat16) {
    launch_embedding_ln<bf16>(
        ids.data_ptr<long>(), pos_ids.data_ptr<long>(),
        type_ids.data_ptr<long>(), (const bf16*)word_w.data_ptr(),
        (const bf16*)pos_w.data_ptr(), (const bf16*)type_w.data_ptr(),
        (const bf16*)ln_w.data_ptr(), (const bf16*)ln_b.data_ptr(),
        (bf16*)y.data_ptr(), tokens, dim, (float)eps, cur_stream());
  } else if (word_w.scalar_type() == at::kFloat) {
    launch_embedding_ln<float>(
        ids.data_ptr<long>(), pos_ids.data_ptr<long>(),
        type_ids.data_ptr<long>(), word_w.data_ptr<float>(),
        pos_w.data_ptr<float>(), type_w.data_ptr<float>(),
        ln_w.data_ptr<float>(), ln_b.data_ptr<float>(),
        y.data_ptr<float>(), tokens, dim, (float)eps, cur_stream());
  } else {
    TORCH_CHECK(false, "embedding_ln: dtype must be bf16/f32");
  }
  return y;
}

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({16, 32}) &&
              b.sizes() == at::IntArrayRef({32, 16}));
  auto c = at::empty({16, 16}, a.options().dtype(at::kFloat));
  launch_mfma_probe((const bf16*)a.contiguous().data_ptr(),
                    (const bf16*)b.contiguous().data_ptr(),
                    c.data_ptr<float>(), cur_stream());
  return c;
}

at::Tensor bperm_probe(const at::Tensor& idx, const at::Tensor& src) {
  TORCH_CHECK(idx.numel() == 64 && src.numel() == 64, "bperm_probe: 64 lanes");
  auto out = at::empty({64}, idx.options());
  launch_bperm_probe((const int*)idx.contiguous().data_ptr(),
                     (const int*)src.contiguous().data_ptr(),
                     (int*)out.data_ptr(),
                     cur_stream());
  return out;
}

at::Tensor tr16_probe(const at::Tensor& dummy, int64_t mode) {
  auto out = at::empty({64, 4}, dummy.options().dtype(at::kShort));
  launch_tr16_probe((short*)out.data_ptr(), (int)mode, cur_stream());
  return out;
}

// hipBLASLt workspaces: ONE persistent buffer per stream, never freed.
// A per-call at::empty workspace is freed while the async GEMM still uses
// it; with dual-stream hipGraph capture the allocator can hand the same
// block to the other stream's GEMM -> replay-time race (measured as a
// stable 2e-2 drift between replays). Keyed by stream so concurrent
// streams never share.
static void* ws_for_stream(hipStream_t stream, const at::TensorOptions& opts,
                           size_t bytes) {
  static std::mutex mu;
  static std::map<hipStream_t, at::Tensor> cache;
  std::lock_guard<std::mutex> lock(mu);
  auto it = cache.find(stream);
  if (it == cache.end()) {
    it = cache.emplace(stream, at::empty({(long)bytes}, opts.dtype(at::kByte))).first;
  }
  return it->second.data_ptr();
}

at::Tensor linear_bias_gelu(const at::Tensor& x, const at::Tensor& w,
                            const at::Tensor& bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16, "linear_bias_gelu: bf16");
  TORCH_CHECK(w.dim() == 2 && bias.dim() == 1 && w.size(0) == bias.size(0));
  auto xc = x.contiguous();
  const long K = x.size(-1), N = w.size(0);
  const long M = x.numel() / K;
  TORCH_CHECK(w.size(1) == K);
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x.options());
  // 256 MB (was 32): hipBLASLt's heuristic can select algos whose real
  // workspace need exceeds the preference cap at large M — a 32 MB
  // workspace then overruns (standalone: "write access to a read-only
  // page"; inside the allocator pool: silent scribble). See the r2 fault
  // isolation in profiles/r07 §3.
  constexpr size_t kWs = 256L * 1024 * 1024;
  hipStream_t stream = cur_stream();
  void* ws = ws_for_stream(stream, x.options(), kWs);
  int rc = hipblaslt_linear_gelu(
      xc.data_ptr(), w.contiguous().data_ptr(), bias.contiguous().data_ptr(),
      y.data_ptr(), M, N, K, ws, kWs, stream);
  TORCH_CHECK(rc == 0, "hipblaslt_linear_gelu failed (no algo)");
  return y;
}

std::tuple<at::Tensor, at::Tensor> residual_layer_norm_fp8(
    const at::Tensor& x, const c10::optional<at::Tensor>& res,
    const at::Tensor& w, const at::Tensor& b, double eps,
    const at::Tensor& scales, const at::Tensor& amaxes, int64_t site) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  const int dim = (int)x.size(-1);
  const long rows = x.numel() / dim;
  auto y = at::empty_like(x);
  auto y8 = at::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  const bool hr = res.has_value();
  launch_residual_ln_fp8(
      (const bf16*)x.data_ptr(), hr ? (const bf16*)res->data_ptr() : nullptr,
      (const bf16*)w.data_ptr(), (const bf16*)b.data_ptr(), (bf16*)y.data_ptr(),
      rows, dim, (float)eps, (unsigned char*)y8.data_ptr(),
      scales.data_ptr<float>(), amaxes.data_ptr<float>(), (int)site,
      cur_stream());
  return {y, y8};
}

std::tuple<at::Tensor, at::Tensor> bias_gelu_fp8(
    const at::Tensor& x, const c10::optional<at::Tensor>& bias,
    const at::Tensor& scales, const at::Tensor& amaxes, int64_t site) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  const int dim = (int)x.size(-1);
  const long n = x.numel();
  auto y = at::empty_like(x);
  auto y8 = at::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  const bool hb = bias.has_value();
  launch_bias_gelu_fp8((const bf16*)x.data_ptr(),
                       hb ? (const bf16*)bias->data_ptr() : nullptr,
                       (bf16*)y.data_ptr(), n, dim,
                       (unsigned char*)y8.data_ptr(), scales.data_ptr<float>(),
                       amaxes.data_ptr<float>(), (int)site, cur_stream());
  return {y, y8};
}

at::Tensor quantize_fp8(const at::Tensor& x, const at::Tensor& scales,
                        const at::Tensor& amaxes, int64_t site) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  auto xc = x.contiguous();
  auto y8 = at::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  launch_quantize_fp8((const bf16*)xc.data_ptr(), (unsigned char*)y8.data_ptr(),
                      scales.data_ptr<float>(), amaxes.data_ptr<float>(),
                      (int)site, x.numel(), cur_stream());
  return y8;
}

void update_fp8_scales(at::Tensor& scales, at::Tensor& inv_scales,
                       at::Tensor& amaxes) {
  TORCH_CHECK(scales.is_cuda() && scales.scalar_type() == at::kFloat);
  launch_update_fp8_scales(scales.data_ptr<float>(),
                           inv_scales.data_ptr<float>(),
                           amaxes.data_ptr<float>(), (int)scales.numel(),
                           cur_stream());
}

at::Tensor fp8_linear(const at::Tensor& x8, const at::Tensor& w8,
                      const at::Tensor& bias, const at::Tensor& w_scale,
                      const at::Tensor& x_scale) {
  TORCH_CHECK(x8.is_cuda() && x8.scalar_type() == at::kFloat8_e4m3fn);
  const long K = x8.size(-1), N = w8.size(0);
  const long M = x8.numel() / K;
  auto sizes = x8.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x8.options().dtype(at::kBFloat16));
  // 256 MB (was 32): hipBLASLt's heuristic can select algos whose real
  // workspace need exceeds the preference cap at large M — a 32 MB
  // workspace then overruns (standalone: "write access to a read-only
  // page"; inside the allocator pool: silent scribble). See the r2 fault
  // isolation in profiles/r07 §3.
  constexpr size_t kWs = 256L * 1024 * 1024;
  hipStream_t stream = cur_stream();
  void* ws = ws_for_stream(stream, x8.options(), kWs);
  int rc = hipblaslt_fp8_linear(
      x8.data_ptr(), w8.data_ptr(), bias.data_ptr(), w_scale.data_ptr(),
      x_scale.data_ptr(), y.data_ptr(), M, N, K, ws, kWs, stream);
  TORCH_CHECK(rc == 0, "hipblaslt_fp8_linear: no algo");
  return y;
}

at::Tensor fp8_linear_gelu_fp8out(const at::Tensor& x8, const at::Tensor& w8,
                                  const at::Tensor& bias,
                                  const at::Tensor& w_scale,
                                  const at::Tensor& x_scale,
                                  const at::Tensor& d_inv_scale,
                                  at::Tensor& amax_out) {
  TORCH_CHECK(x8.is_cuda() && x8.scalar_type() == at::kFloat8_e4m3fn);
  const long K = x8.size(-1), N = w8.size(0);
  const long M = x8.numel() / K;
  auto sizes = x8.sizes().vec();
  sizes.back() = N;
  auto y8 = at::empty(sizes, x8.options());
  // 256 MB (was 32): hipBLASLt's heuristic can select algos whose real
  // workspace need exceeds the preference cap at large M — a 32 MB
  // workspace then overruns (standalone: "write access to a read-only
  // page"; inside the allocator pool: silent scribble). See the r2 fault
  // isolation in profiles/r07 §3.
  constexpr size_t kWs = 256L * 1024 * 1024;
  hipStream_t stream = cur_stream();
  void* ws = ws_for_stream(stream, x8.options(), kWs);
  int rc = hipblaslt_fp8_linear_gelu_fp8out(
      x8.data_ptr(), w8.data_ptr(), bias.data_ptr(), w_scale.data_ptr(),
      x_scale.data_ptr(), d_inv_scale.data_ptr(), amax_out.data_ptr(),
      y8.data_ptr(), M, N, K, ws, kWs, stream);
  TORCH_CHECK(rc == 0, "hipblaslt_fp8_linear_gelu_fp8out: no algo");
  return y8;
}

at::Tensor linear_bias(const at::Tensor& x, const at::Tensor& w,
                       const at::Tensor& bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16, "linear_bias: bf16");
  TORCH_CHECK(w.dim() == 2 && bias.dim() == 1 && w.size(0) == bias.size(0));
  auto xc = x.contiguous();
  const long K = x.size(-1), N = w.size(0);
  const long M = x.numel() / K;
  TORCH_CHECK(w.size(1) == K);
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x.options());
  // 256 MB (was 32): hipBLASLt's heuristic can select algos whose real
  // workspace need exceeds the preference cap at large M — a 32 MB
  // workspace then overruns (standalone: "write access to a read-only
  // page"; inside the allocator pool: silent scribble). See the r2 fault
  // isolation in profiles/r07 §3.
  constexpr size_t kWs = 256L * 1024 * 1024;
  hipStream_t stream = cur_stream();
  void* ws = ws_for_stream(stream, x.options(), kWs);
  int rc = hipblaslt_linear_bias(
      xc.data_ptr(), w.contiguous().data_ptr(), bias.contiguous().data_ptr(),
      y.data_ptr(), M, N, K, ws, kWs, stream);
  TORCH_CHECK(rc == 0, "hipblaslt_linear_bias failed (no algo)");
  return y;
}

at::Tensor linear_bias_residual(const at::Tensor& x, const at::Tensor& w,
                                const at::Tensor& bias,
                                const at::Tensor& residual) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16, "linear_bias_residual: bf16");
  TORCH_CHECK(w.dim() == 2 && bias.dim() == 1 && w.size(0) == bias.size(0));
  auto xc = x.contiguous();
  auto rc = residual.contiguous();
  const long K = x.size(-1), N = w.size(0);
  const long M = x.numel() / K;
  TORCH_CHECK(w.size(1) == K && rc.numel() == M * N);
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x.options());
  // 256 MB (was 32): hipBLASLt's heuristic can select algos whose real
  // workspace need exceeds the preference cap at large M — a 32 MB
  // workspace then overruns (standalone: "write access to a read-only
  // page"; inside the allocator pool: silent scribble). See the r2 fault
  // isolation in profiles/r07 §3.
  constexpr size_t kWs = 256L * 1024 * 1024;
  hipStream_t stream = cur_stream();
  void* ws = ws_for_stream(stream, x.options(), kWs);
  int rc2 = hipblaslt_linear_bias_add(
      xc.data_ptr(), w.contiguous().data_ptr(), bias.contiguous().data_ptr(),
      rc.data_ptr(), y.data_ptr(), M, N, K, ws, kWs, stream);
  TORCH_CHECK(rc2 == 0, "hipblaslt_linear_bias_add failed (no algo)");
  return y;
}

at::Tensor roi_align(const at::Tensor& input, const at::Tensor& rois,
                     int64_t ph, int64_t pw, double spatial_scale,
                     int64_t sampling_ratio) {
  TORCH_CHECK(input.is_cuda() && input.dim() == 4 && input.is_contiguous());
  TORCH_CHECK(rois.is_cuda() && rois.dim() == 2 && rois.size(1) == 5);
  TORCH_CHECK(rois.scalar_type() == at::kFloat);
  const int N = (int)input.size(0), C = (int)input.size(1),
            H = (int)input.size(2), W = (int)input.size(3);
  const int R = (int)rois.size(0);
  auto out = at::empty({R, (long)C, ph, pw}, input.options());
  if (input.scalar_type() == at::kFloat) {
    launch_roi_align<float>(input.data_ptr<float>(),
                            rois.contiguous().data_ptr<float>(),
                            out.data_ptr<float>(), N, C, H, W, R, (int)ph,
                            (int)pw, (float)spatial_scale, (int)sampling_ratio,
                            cur_stream());
  } else if (input.scalar_type() == at::kBFloat16) {
    launch_roi_align<bf16>((const bf16*)input.data_ptr(),
                           rois.contiguous().data_ptr<float>(),
                           (bf16*)out.data_ptr(), N, C, H, W, R, (int)ph,
                           (int)pw, (float)spatial_scale, (int)sampling_ratio,
                           cur_stream());
  } else {
    TORCH_CHECK(false, "roi_align: dtype must be f32/bf16");
  }
  return out;
}

at::Tensor nms_multiclass(const at::Tensor& boxes, const at::Tensor& scores,
                          double iou_thr, double score_thr) {
  TORCH_CHECK(boxes.is_cuda() && boxes.scalar_type() == at::kFloat);
  TORCH_CHECK(scores.is_cuda() && scores.scalar_type() == at::kFloat);
  const int R = (int)boxes.size(0);
  const int C = (int)scores.size(1);
  TORCH_CHECK(R <= 1024, "nms_multiclass: R <= 1024");
  auto order = std::get<1>(scores.sort(0, /*descending=*/true)).contiguous();
  auto out = at::zeros_like(scores);
  launch_nms_multiclass(boxes.contiguous().data_ptr<float>(),
                        scores.contiguous().data_ptr<float>(),
                        order.data_ptr<long>(), out.data_ptr<float>(),
                        R, C, (float)iou_thr, (float)score_thr, cur_stream());
  return out;
}

// ---- training-path fused LN / GELU backward (train_bwd.hip) ---------------

constexpr int kBwdChunks = 128;  // fixed partial-chunk count => deterministic

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> ln_fwd_train(
    const at::Tensor& x, const c10::optional<at::Tensor>& res,
    const at::Tensor& w, const at::Tensor& b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                  x.scalar_type() == at::kBFloat16,
              "ln_fwd_train: x bf16 contiguous");
  const int dim = (int)x.size(-1);
  const long rows = x.numel() / dim;
  auto y = at::empty_like(x);
  const bool hr = res.has_value();
  if (hr) TORCH_CHECK(res->is_contiguous() && res->sizes() == x.sizes());
  // xs = x + res (saved for backward); without a residual x itself is xs
  auto xs = hr ? at::empty_like(x) : x;
  auto f32 = x.options().dtype(at::kFloat);
  auto mean = at::empty({rows}, f32);
  auto rstd = at::empty({rows}, f32);
  launch_ln_fwd_train((const bf16*)x.data_ptr(),
                      hr ? (const bf16*)res->data_ptr() : nullptr,
                      (const bf16*)w.data_ptr(), (const bf16*)b.data_ptr(),
                      (bf16*)y.data_ptr(), (bf16*)xs.data_ptr(),
                      mean.data_ptr<float>(), rstd.data_ptr<float>(), rows,
                      dim, (float)eps, cur_stream());
  return {y, xs, mean, rstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_bwd(
    const at::Tensor& gy, const at::Tensor& xs, const at::Tensor& mean,
    const at::Tensor& rstd, const at::Tensor& w) {
  TORCH_CHECK(gy.is_cuda() && gy.is_contiguous() &&
                  gy.scalar_type() == at::kBFloat16,
              "ln_bwd: grad bf16 contiguous");
  const int dim = (int)gy.size(-1);
  const long rows = gy.numel() / dim;
  auto gxs = at::empty_like(gy);
  launch_ln_bwd_input((const bf16*)gy.data_ptr(), (const bf16*)xs.data_ptr(),
                      mean.data_ptr<float>(), rstd.data_ptr<float>(),
                      (const bf16*)w.data_ptr(), (bf16*)gxs.data_ptr(), rows,
                      dim, cur_stream());
  const int nchunks = (int)std::min<long>(kBwdChunks, rows);
  const int rpc = (int)((rows + nchunks - 1) / nchunks);
  auto f32 = gy.options().dtype(at::kFloat);
  auto gw_part = at::empty({nchunks, dim}, f32);
  auto gb_part = at::empty({nchunks, dim}, f32);
  launch_ln_bwd_param((const bf16*)gy.data_ptr(), (const bf16*)xs.data_ptr(),
                      mean.data_ptr<float>(), rstd.data_ptr<float>(),
                      gw_part.data_ptr<float>(), gb_part.data_ptr<float>(),
                      rows, dim, nchunks, rpc, cur_stream());
  return {gxs, gw_part.sum(0).to(w.scalar_type()),
          gb_part.sum(0).to(w.scalar_type())};
}

std::tuple<at::Tensor, at::Tensor> bias_gelu_bwd(const at::Tensor& gy,
                                                 const at::Tensor& pre) {
  TORCH_CHECK(gy.is_cuda() && gy.is_contiguous() &&
                  gy.scalar_type() == at::kBFloat16 &&
                  pre.is_contiguous() && pre.sizes() == gy.sizes(),
              "bias_gelu_bwd: bf16 contiguous");
  const int dim = (int)gy.size(-1);
  const long rows = gy.numel() / dim;
  auto gpre = at::empty_like(gy);
  const int nchunks = (int)std::min<long>(kBwdChunks, rows);
  const int rpc = (int)((rows + nchunks - 1) / nchunks);
  auto gb_part = at::empty({nchunks, dim}, gy.options().dtype(at::kFloat));
  launch_bias_gelu_bwd((const bf16*)gy.data_ptr(), (const bf16*)pre.data_ptr(),
                       (bf16*)gpre.data_ptr(), gb_part.data_ptr<float>(), rows,
                       dim, nchunks, rpc, cur_stream());
  return {gpre, gb_part.sum(0).to(gy.scalar_type())};
}

}  // namespace

TORCH_LIBRARY(vilbert_amd, m) {
  m.def("residual_layer_norm(Tensor x, Tensor? res, Tensor w, Tensor b, float eps) -> Tensor");
  m.def("bias_gelu(Tensor x, Tensor? bias) -> Tensor");
  m.def("attention(Tensor q, Tensor k, Tensor v, int heads, Tensor? mask) -> Tensor");
  m.def("adamw_step(Tensor(a!) param, Tensor grad, Tensor(b!) master, Tensor(c!) m, Tensor(d!) v, float lr, float b1, float b2, float eps, float wd, int step) -> ()");
  m.def("mfma_linear(Tensor x, Tensor w, Tensor? bias, Tensor? residual, bool gelu) -> Tensor");
  m.def("attention_train_fwd(Tensor q, Tensor k, Tensor v, int heads, Tensor? mask, Tensor? dropm) -> (Tensor, Tensor)");
  m.def("attention_bwd(Tensor q, Tensor k, Tensor v, Tensor probs, Tensor? dropm, Tensor dout, int heads) -> (Tensor, Tensor, Tensor)");
  m.def("attention_probs(Tensor q, Tensor k, Tensor v, int heads, Tensor? mask) -> (Tensor, Tensor)");
  m.def("attention_fp8out(Tensor q, Tensor k, Tensor v, int heads, Tensor? mask, Tensor scales, Tensor(a!) amaxes, int site) -> (Tensor, Tensor)");
  m.def("embedding_ln(Tensor ids, Tensor pos, Tensor type, Tensor word_w, Tensor pos_w, Tensor type_w, Tensor ln_w, Tensor ln_b, float eps) -> Tensor");
  m.def("mfma_probe(Tensor a, Tensor b) -> Tensor");
  m.def("bperm_probe(Tensor idx, Tensor src) -> Tensor");
  m.def("tr16_probe(Tensor dummy, int mode) -> Tensor");
  m.def("nms_multiclass(Tensor boxes, Tensor scores, float iou_thr, float score_thr) -> Tensor");
  m.def("roi_align(Tensor input, Tensor rois, int ph, int pw, float spatial_scale, int sampling_ratio) -> Tensor");
  m.def("linear_bias_gelu(Tensor x, Tensor w, Tensor bias) -> Tensor");
  m.def("linear_bias_residual(Tensor x, Tensor w, Tensor bias, Tensor residual) -> Tensor");
  m.def("linear_bias(Tensor x, Tensor w, Tensor bias) -> Tensor");
  m.def("residual_layer_norm_fp8(Tensor x, Tensor? res, Tensor w, Tensor b, float eps, Tensor scales, Tensor amaxes, int site) -> (Tensor, Tensor)");
  m.def("bias_gelu_fp8(Tensor x, Tensor? bias, Tensor scales, Tensor amaxes, int site) -> (Tensor, Tensor)");
  m.def("quantize_fp8(Tensor x, Tensor scales, Tensor amaxes, int site) -> Tensor");
  m.def("update_fp8_scales(Tensor(a!) scales, Tensor(b!) inv_scales, Tensor(c!) amaxes) -> ()");
  m.def("fp8_linear(Tensor x8, Tensor w8, Tensor bias, Tensor w_scale, Tensor x_scale) -> Tensor");
  m.def("fp8_linear_gelu_fp8out(Tensor x8, Tensor w8, Tensor bias, Tensor w_scale, Tensor x_scale, Tensor d_inv_scale, Tensor(a!) amax_out) -> Tensor");
  m.def("ln_fwd_train(Tensor x, Tensor? res, Tensor w, Tensor b, float eps) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("ln_bwd(Tensor gy, Tensor xs, Tensor mean, Tensor rstd, Tensor w) -> (Tensor, Tensor, Tensor)");
  m.def("bias_gelu_bwd(Tensor gy, Tensor pre) -> (Tensor, Tensor)");
}

TORCH_LIBRARY_IMPL(vilbert_amd, CUDA, m) {
  m.impl("residual_layer_norm", residual_layer_norm);
  m.impl("bias_gelu", bias_gelu);
  m.impl("attention", attention);
  m.impl("attention_probs", attention_probs);
  m.impl("attention_train_fwd", attention_train_fwd);
  m.impl("attention_bwd", attention_bwd);
  m.impl("mfma_linear", mfma_linear);
  m.impl("adamw_step", adamw_step);
  m.impl("attention_fp8out", attention_fp8out);
  m.impl("embedding_ln", embedding_ln);
  m.impl("mfma_probe", mfma_probe);
  m.impl("bperm_probe", bperm_probe);
  m.impl("tr16_probe", tr16_probe);
  m.impl("nms_multiclass", nms_multiclass);
  m.impl("roi_align", roi_align);
  m.impl("linear_bias_gelu", linear_bias_gelu);
  m.impl("linear_bias_residual", linear_bias_residual);
  m.impl("linear_bias", linear_bias);
  m.impl("residual_layer_norm_fp8", residual_layer_norm_fp8);
  m.impl("bias_gelu_fp8", bias_gelu_fp8);
  m.impl("quantize_fp8", quantize_fp8);
  m.impl("update_fp8_scales", update_fp8_scales);
  m.impl("fp8_linear", fp8_linear);
  m.impl("fp8_linear_gelu_fp8out", fp8_linear_gelu_fp8out);
  m.impl("ln_fwd_train", ln_fwd_train);
  m.impl("ln_bwd", ln_bwd);
  m.impl("bias_gelu_bwd", bias_gelu_bwd);
}
