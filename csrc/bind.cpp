// Python bindings for the tiny_deepspeed_amd CDNA4 kernel extension (_C).
//
// Host-side glue only: shape/dtype checks, output allocation, stream lookup.
// All device code lives in csrc/kernels/*.hip (pure HIP, no torch headers).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <hip/hip_runtime_api.h>

#include <array>
#include <cstring>
#include <vector>

// ---- extern "C" launcher prototypes (csrc/kernels/*.hip) ------------------
extern "C" {
hipError_t tdsa_ln_fwd(const void*, const void*, void*, const void*,
                       const void*, void*, float*, float*, int, int, float,
                       int, hipStream_t);
int tdsa_ln_bwd_dx_stripes(int M);
hipError_t tdsa_ln_bwd_dx(const void*, const void*, const void*, const void*,
                          const float*, const float*, void*, float*, float*,
                          int, int, int, hipStream_t);
hipError_t tdsa_ln_bwd_dwdb(const float*, const float*, float*, float*, int,
                            int, hipStream_t);
hipError_t tdsa_gelu_fwd(const void*, void*, long long, int, hipStream_t);
hipError_t tdsa_gelu_bwd(const void*, const void*, void*, long long, int,
                         hipStream_t);
hipError_t tdsa_column_sum(const void*, float*, void*, int, int, int,
                           hipStream_t);
hipError_t tdsa_embedding_fwd(const void*, const long long*, void*, long long,
                              int, int, hipStream_t);
hipError_t tdsa_embedding_bwd(const void*, const long long*, float*, long long,
                              int, long long, int, hipStream_t);
hipError_t tdsa_ce_fwd(const void*, const long long*, float*, float*, int*,
                       long long, int, long long, int, hipStream_t);
hipError_t tdsa_ce_bwd(const void*, const long long*, const float*, void*,
                       long long, int, float, long long, int, hipStream_t);
hipError_t tdsa_adamw_step(void*, const void*, float*, float*, float*, float*,
                           int, int, float, float, float, float, float,
                           long long, long long, int, int, hipStream_t);
hipError_t tdsa_adamw_multi(const void*, long long, int, int, float, float,
                            float, float, float, long long, hipStream_t);
int tdsa_adamw_desc_size();
int tdsa_adamw_chunkref_size();
int tdsa_sgd_desc_size();
hipError_t tdsa_sgd_multi(const void*, long long, int, int, float, float,
                          float, float, int, int, int, hipStream_t);
hipError_t tdsa_sgd_step(void*, const void*, float*, float*, int, int, float,
                         float, float, float, int, int, int, long long, int,
                         int, hipStream_t);
hipError_t tdsa_attn_fwd(const void*, const void*, const void*, void*, float*,
                         long long, long long, int, float, const long long*,
                         const long long*, float, unsigned long long,
                         hipStream_t);
hipError_t tdsa_attn_bwd(const void*, const void*, const void*, const void*,
                         const float*, const void*, void*, void*, void*, float*,
                         long long, long long, int, float, const long long*,
                         const long long*, const long long*, float,
                         unsigned long long, hipStream_t);
int tdsa_gemm_tn_splits(long long M, int N, int K);
hipError_t tdsa_gemm_tn(const void*, const void*, float*, long long, int, int,
                        hipStream_t);
hipError_t tdsa_dbg_mfma(const void*, const void*, float*, int, hipStream_t);
hipError_t tdsa_dbg_mfma32(const void*, const void*, float*, hipStream_t);
hipError_t tdsa_dbg_stage(const void*, void*, int, hipStream_t);
hipError_t tdsa_dbg_tr16(const void*, float*, hipStream_t);
}

namespace {

void check_hip(hipError_t err, const char* what) {
  TORCH_CHECK(err == hipSuccess, what, ": ", hipGetErrorString(err));
}

int dtype_flag(const at::Tensor& t) {
  if (t.scalar_type() == at::kBFloat16) return 1;
  if (t.scalar_type() == at::kFloat) return 0;
  TORCH_CHECK(false, "expected float32 or bfloat16, got ", t.scalar_type());
  return -1;
}

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

#define CHECK_IN(t)                                                     \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                     \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// ---- layernorm ------------------------------------------------------------
std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                                      double eps,
                                      c10::optional<at::Tensor> res) {
  CHECK_IN(x); CHECK_IN(w); CHECK_IN(b);
  const int N = x.size(-1);
  const long long M = x.numel() / N;
  auto y = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  auto mean = at::empty({M}, f32);
  auto rstd = at::empty({M}, f32);
  at::Tensor h;
  void* res_p = nullptr;
  void* h_p = nullptr;
  if (res.has_value()) {
    CHECK_IN((*res));
    h = at::empty_like(x);
    res_p = res->data_ptr();
    h_p = h.data_ptr();
  }
  check_hip(tdsa_ln_fwd(x.data_ptr(), res_p, h_p, w.data_ptr(), b.data_ptr(),
                        y.data_ptr(), mean.data_ptr<float>(),
                        rstd.data_ptr<float>(), (int)M, N, (float)eps,
                        dtype_flag(x), cur_stream()),
            "layernorm_fwd");
  // mean/rstd viewed to the row shape of x
  auto row_sizes = x.sizes().vec();
  row_sizes.pop_back();
  if (res.has_value())
    return {y, mean.view(row_sizes), rstd.view(row_sizes), h};
  return {y, mean.view(row_sizes), rstd.view(row_sizes)};
}

std::vector<at::Tensor> layernorm_bwd_dx(at::Tensor dy, at::Tensor x,
                                         at::Tensor w, at::Tensor mean,
                                         at::Tensor rstd, int64_t n_stripes,
                                         c10::optional<at::Tensor> dh) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(w);
  (void)n_stripes;  // stripe count is chosen device-side for full occupancy
  const int N = x.size(-1);
  const long long M = x.numel() / N;
  const int G = tdsa_ln_bwd_dx_stripes((int)M);
  auto dx = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  auto pdw = at::empty({G, N}, f32);
  auto pdb = at::empty({G, N}, f32);
  auto meanc = mean.contiguous();
  auto rstdc = rstd.contiguous();
  at::Tensor dhc;
  void* dh_p = nullptr;
  if (dh.has_value()) {
    dhc = dh->contiguous();
    dh_p = dhc.data_ptr();
  }
  check_hip(tdsa_ln_bwd_dx(dy.data_ptr(), dh_p, x.data_ptr(), w.data_ptr(),
                           meanc.data_ptr<float>(), rstdc.data_ptr<float>(),
                           dx.data_ptr(), pdw.data_ptr<float>(),
                           pdb.data_ptr<float>(), (int)M, N, dtype_flag(x),
                           cur_stream()),
            "layernorm_bwd_dx");
  return {dx, pdw, pdb};
}

std::vector<at::Tensor> layernorm_bwd_dwdb(at::Tensor pdw, at::Tensor pdb) {
  CHECK_IN(pdw); CHECK_IN(pdb);
  const int G = pdw.size(0);
  const int N = pdw.size(1);
  auto dw = at::empty({N}, pdw.options());
  auto db = at::empty({N}, pdb.options());
  check_hip(tdsa_ln_bwd_dwdb(pdw.data_ptr<float>(), pdb.data_ptr<float>(),
                             dw.data_ptr<float>(), db.data_ptr<float>(), G, N,
                             cur_stream()),
            "layernorm_bwd_dwdb");
  return {dw, db};
}

// ---- elementwise ----------------------------------------------------------
at::Tensor gelu_fwd(at::Tensor x) {
  CHECK_IN(x);
  auto y = at::empty_like(x);
  check_hip(tdsa_gelu_fwd(x.data_ptr(), y.data_ptr(), x.numel(), dtype_flag(x),
                          cur_stream()),
            "gelu_fwd");
  return y;
}

at::Tensor gelu_bwd(at::Tensor dy, at::Tensor x) {
  CHECK_IN(dy); CHECK_IN(x);
  auto dx = at::empty_like(x);
  check_hip(tdsa_gelu_bwd(dy.data_ptr(), x.data_ptr(), dx.data_ptr(), x.numel(),
                          dtype_flag(x), cur_stream()),
            "gelu_bwd");
  return dx;
}

at::Tensor column_sum(at::Tensor dy) {
  CHECK_IN(dy);
  TORCH_CHECK(dy.dim() == 2, "column_sum expects 2-D input");
  const int M = dy.size(0);
  const int N = dy.size(1);
  auto out32 = at::zeros({N}, dy.options().dtype(at::kFloat));
  auto out = at::empty({N}, dy.options());
  check_hip(tdsa_column_sum(dy.data_ptr(), out32.data_ptr<float>(),
                            out.data_ptr(), M, N, dtype_flag(dy), cur_stream()),
            "column_sum");
  return out;
}

// ---- embedding ------------------------------------------------------------
at::Tensor embedding_fwd(at::Tensor weight, at::Tensor idx) {
  CHECK_IN(weight); CHECK_IN(idx);
  TORCH_CHECK(idx.scalar_type() == at::kLong, "idx must be int64");
  const long long R = idx.numel();
  const int D = weight.size(1);
  auto out = at::empty({R, (long long)D}, weight.options());
  check_hip(tdsa_embedding_fwd(weight.data_ptr(), (const long long*)idx.data_ptr<int64_t>(),
                               out.data_ptr(), R, D, dtype_flag(weight),
                               cur_stream()),
            "embedding_fwd");
  return out;
}

at::Tensor embedding_bwd(at::Tensor dy, at::Tensor idx, int64_t num_embeddings,
                         int64_t padding_idx) {
  CHECK_IN(dy); CHECK_IN(idx);
  TORCH_CHECK(idx.scalar_type() == at::kLong, "idx must be int64");
  const long long R = idx.numel();
  const int D = dy.size(-1);
  auto dw32 = at::zeros({num_embeddings, (long long)D},
                        dy.options().dtype(at::kFloat));
  check_hip(tdsa_embedding_bwd(dy.data_ptr(), (const long long*)idx.data_ptr<int64_t>(),
                               dw32.data_ptr<float>(), R, D, padding_idx,
                               dtype_flag(dy), cur_stream()),
            "embedding_bwd");
  return dw32;
}

// ---- cross entropy --------------------------------------------------------
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits, at::Tensor targets,
                                          int64_t ignore_index) {
  CHECK_IN(logits); CHECK_IN(targets);
  TORCH_CHECK(logits.dim() == 2, "logits must be 2-D");
  TORCH_CHECK(targets.scalar_type() == at::kLong, "targets must be int64");
  const long long R = logits.size(0);
  const int V = logits.size(1);
  auto f32 = logits.options().dtype(at::kFloat);
  auto lse = at::empty({R}, f32);
  auto loss_sum = at::zeros({}, f32);
  auto n_valid32 = at::zeros({}, logits.options().dtype(at::kInt));
  check_hip(tdsa_ce_fwd(logits.data_ptr(), (const long long*)targets.data_ptr<int64_t>(),
                        lse.data_ptr<float>(), loss_sum.data_ptr<float>(),
                        n_valid32.data_ptr<int>(), R, V, ignore_index,
                        dtype_flag(logits), cur_stream()),
            "cross_entropy_fwd");
  return {loss_sum, lse, n_valid32.to(at::kLong)};
}

at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor targets,
                             at::Tensor lse, double dloss, int64_t n_valid,
                             int64_t ignore_index,
                             c10::optional<at::Tensor> out) {
  CHECK_IN(logits); CHECK_IN(targets); CHECK_IN(lse);
  const long long R = logits.size(0);
  const int V = logits.size(1);
  at::Tensor dlogits;
  if (out.has_value()) {  // e.g. a row-slice of the fused-CE dlogits buffer
    dlogits = *out;
    CHECK_IN(dlogits);
    TORCH_CHECK(dlogits.sizes() == logits.sizes()
                && dlogits.scalar_type() == logits.scalar_type(),
                "cross_entropy_bwd: out must match logits");
  } else {
    dlogits = at::empty_like(logits);
  }
  const float scale = (float)(dloss / (double)std::max<int64_t>(n_valid, 1));
  check_hip(tdsa_ce_bwd(logits.data_ptr(), (const long long*)targets.data_ptr<int64_t>(),
                        lse.data_ptr<float>(), dlogits.data_ptr(), R, V, scale,
                        ignore_index, dtype_flag(logits), cur_stream()),
            "cross_entropy_bwd");
  return dlogits;
}

// ---- fused optimizers -----------------------------------------------------
void adamw_step(at::Tensor param, at::Tensor grad, at::Tensor m, at::Tensor v,
                at::Tensor master, at::Tensor vmax, bool has_master,
                bool amsgrad, double lr, double b1, double b2, double eps,
                double wd, int64_t step) {
  CHECK_IN(param); CHECK_IN(grad); CHECK_IN(m); CHECK_IN(v);
  check_hip(tdsa_adamw_step(param.data_ptr(), grad.data_ptr(),
                            m.data_ptr<float>(), v.data_ptr<float>(),
                            has_master ? master.data_ptr<float>() : nullptr,
                            amsgrad ? vmax.data_ptr<float>() : nullptr,
                            has_master, amsgrad, (float)lr, (float)b1,
                            (float)b2, (float)eps, (float)wd, step,
                            param.numel(), dtype_flag(param), dtype_flag(grad),
                            cur_stream()),
            "adamw_step");
}

// One fused launch for a whole parameter list (no amsgrad; the python
// optimizer falls back to per-tensor calls for that).
void adamw_step_multi(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                      std::vector<c10::optional<at::Tensor>> masters,
                      double lr, double b1, double b2, double eps, double wd,
                      int64_t step) {
  const size_t n = params.size();
  TORCH_CHECK(grads.size() == n && ms.size() == n && vs.size() == n &&
              masters.size() == n, "length mismatch");
  if (n == 0) return;
  struct Desc {  // must mirror AdamTensorDesc in optim.hip
    void* p; const void* g; float* m; float* v; float* master;
    long long numel; int param_bf16; int grad_bf16;
  };
  struct CRef { int tensor; int chunk; };
  TORCH_CHECK((int)sizeof(Desc) == tdsa_adamw_desc_size());
  TORCH_CHECK((int)sizeof(CRef) == tdsa_adamw_chunkref_size());
  const int chunk_elems = 1 << 16;
  std::vector<Desc> descs(n);
  std::vector<CRef> chunks;
  chunks.reserve(1024);
  for (size_t i = 0; i < n; ++i) {
    auto& p = params[i];
    auto& g = grads[i];
    Desc d;
    d.p = p.data_ptr();
    d.g = g.data_ptr();
    d.m = ms[i].data_ptr<float>();
    d.v = vs[i].data_ptr<float>();
    d.master = masters[i].has_value() ? masters[i]->data_ptr<float>() : nullptr;
    d.numel = p.numel();
    d.param_bf16 = dtype_flag(p);
    d.grad_bf16 = dtype_flag(g);
    descs[i] = d;
    const long long nch = (d.numel + chunk_elems - 1) / chunk_elems;
    for (long long c = 0; c < nch; ++c)
      chunks.push_back({(int)i, (int)c});
  }
  const long long desc_bytes = (long long)(n * sizeof(Desc));
  const long long total = desc_bytes + (long long)(chunks.size() * sizeof(CRef));
  auto host = at::empty({total}, at::TensorOptions().dtype(at::kByte));
  std::memcpy(host.data_ptr(), descs.data(), desc_bytes);
  std::memcpy((char*)host.data_ptr() + desc_bytes, chunks.data(),
              chunks.size() * sizeof(CRef));
  auto dev = host.to(params[0].device());
  check_hip(tdsa_adamw_multi(dev.data_ptr(), desc_bytes, (int)chunks.size(),
                             chunk_elems, (float)lr, (float)b1, (float)b2,
                             (float)eps, (float)wd, step, cur_stream()),
            "adamw_step_multi");
}

void sgd_step_multi(std::vector<at::Tensor> params,
                    std::vector<at::Tensor> grads,
                    std::vector<c10::optional<at::Tensor>> bufs,
                    std::vector<c10::optional<at::Tensor>> masters,
                    double lr, double momentum, double dampening, double wd,
                    bool nesterov, bool maximize, bool first_step) {
  const size_t n = params.size();
  TORCH_CHECK(grads.size() == n && bufs.size() == n && masters.size() == n,
              "length mismatch");
  if (n == 0) return;
  struct Desc {  // must mirror SgdTensorDesc in optim.hip
    void* p; const void* g; float* buf; float* master;
    long long numel; int param_bf16; int grad_bf16;
  };
  struct CRef { int tensor; int chunk; };
  TORCH_CHECK((int)sizeof(Desc) == tdsa_sgd_desc_size());
  const int chunk_elems = 1 << 16;
  std::vector<Desc> descs(n);
  std::vector<CRef> chunks;
  for (size_t i = 0; i < n; ++i) {
    Desc d;
    d.p = params[i].data_ptr();
    d.g = grads[i].data_ptr();
    d.buf = bufs[i].has_value() ? bufs[i]->data_ptr<float>() : nullptr;
    d.master = masters[i].has_value() ? masters[i]->data_ptr<float>() : nullptr;
    d.numel = params[i].numel();
    d.param_bf16 = dtype_flag(params[i]);
    d.grad_bf16 = dtype_flag(grads[i]);
    descs[i] = d;
    const long long nch = (d.numel + chunk_elems - 1) / chunk_elems;
    for (long long c = 0; c < nch; ++c) chunks.push_back({(int)i, (int)c});
  }
  const long long desc_bytes = (long long)(n * sizeof(Desc));
  const long long total = desc_bytes + (long long)(chunks.size() * sizeof(CRef));
  auto host = at::empty({total}, at::TensorOptions().dtype(at::kByte));
  std::memcpy(host.data_ptr(), descs.data(), desc_bytes);
  std::memcpy((char*)host.data_ptr() + desc_bytes, chunks.data(),
              chunks.size() * sizeof(CRef));
  auto dev = host.to(params[0].device());
  check_hip(tdsa_sgd_multi(dev.data_ptr(), desc_bytes, (int)chunks.size(),
                           chunk_elems, (float)lr, (float)momentum,
                           (float)dampening, (float)wd, nesterov, maximize,
                           first_step, cur_stream()),
            "sgd_step_multi");
}

void sgd_step(at::Tensor param, at::Tensor grad, at::Tensor buf,
              at::Tensor master, bool has_buf, bool has_master, double lr,
              double momentum, double dampening, double wd, bool nesterov,
              bool maximize, bool first_step) {
  CHECK_IN(param); CHECK_IN(grad);
  check_hip(tdsa_sgd_step(param.data_ptr(), grad.data_ptr(),
                          has_buf ? buf.data_ptr<float>() : nullptr,
                          has_master ? master.data_ptr<float>() : nullptr,
                          has_buf, has_master, (float)lr, (float)momentum,
                          (float)dampening, (float)wd, nesterov, maximize,
                          first_step, param.numel(), dtype_flag(param),
                          dtype_flag(grad), cur_stream()),
            "sgd_step");
}

// ---- TN GEMM (linear dW autotuner candidate) ------------------------------
at::Tensor gemm_tn(at::Tensor dy2, at::Tensor x2) {
  CHECK_IN(dy2); CHECK_IN(x2);
  TORCH_CHECK(dy2.dim() == 2 && x2.dim() == 2, "gemm_tn expects 2-D inputs");
  TORCH_CHECK(dy2.scalar_type() == at::kBFloat16
              && x2.scalar_type() == at::kBFloat16, "gemm_tn is bf16-only");
  TORCH_CHECK(dy2.size(0) == x2.size(0), "reduction dims differ");
  const long long M = dy2.size(0);
  const int N = dy2.size(1);
  const int K = x2.size(1);
  const int splits = tdsa_gemm_tn_splits(M, N, K);
  TORCH_CHECK(splits > 0, "gemm_tn: unsupported shape (need M%64==0, "
              "N%128==0, K%128==0), got ", M, "x", N, "/", K);
  auto f32 = dy2.options().dtype(at::kFloat);
  auto dw32 = splits > 1 ? at::zeros({N, K}, f32) : at::empty({N, K}, f32);
  check_hip(tdsa_gemm_tn(dy2.data_ptr(), x2.data_ptr(),
                         dw32.data_ptr<float>(), M, N, K, cur_stream()),
            "gemm_tn");
  return dw32.to(at::kBFloat16);
}

// ---- attention ------------------------------------------------------------
static void check_attn_tensor(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dim() == 4 && t.size(3) == 64, name,
              ": attention kernel requires (B,H,T,64)");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.stride(3) == 1, name, " last dim must be contiguous");
}

static std::array<long long, 3> strides3(const at::Tensor& t) {
  return {(long long)t.stride(0), (long long)t.stride(1),
          (long long)t.stride(2)};
}

std::vector<at::Tensor> attention_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                      double scale,
                                      c10::optional<at::Tensor> out,
                                      double dropout_p, int64_t seed) {
  check_attn_tensor(q, "q");
  check_attn_tensor(k, "k");
  check_attn_tensor(v, "v");
  TORCH_CHECK(q.strides() == k.strides() && q.strides() == v.strides(),
              "q/k/v must share strides");
  TORCH_CHECK(q.size(2) % 64 == 0, "attention kernel requires T %% 64 == 0");
  const long long B = q.size(0), H = q.size(1);
  const int T = q.size(2);
  at::Tensor o = out.has_value() ? *out : at::empty_like(q);
  check_attn_tensor(o, "o");
  auto lse = at::empty({B, H, (long long)T}, q.options().dtype(at::kFloat));
  auto sq = strides3(q);
  auto so = strides3(o);
  TORCH_CHECK(dropout_p >= 0.0 && dropout_p < 1.0, "bad dropout_p");
  check_hip(tdsa_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          o.data_ptr(), lse.data_ptr<float>(), B, H, T,
                          (float)scale, sq.data(), so.data(), (float)dropout_p,
                          (unsigned long long)seed, cur_stream()),
            "attention_fwd");
  return {o, lse};
}

std::vector<at::Tensor> attention_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                      at::Tensor o, at::Tensor lse,
                                      at::Tensor dout, double scale,
                                      c10::optional<at::Tensor> dq_out,
                                      c10::optional<at::Tensor> dk_out,
                                      c10::optional<at::Tensor> dv_out,
                                      double dropout_p, int64_t seed) {
  check_attn_tensor(q, "q");
  check_attn_tensor(k, "k");
  check_attn_tensor(v, "v");
  check_attn_tensor(o, "o");
  check_attn_tensor(dout, "dout");
  TORCH_CHECK(q.strides() == k.strides() && q.strides() == v.strides(),
              "q/k/v must share strides");
  TORCH_CHECK(o.strides() == dout.strides(), "o/dout must share strides");
  auto lsec = lse.contiguous();
  const long long B = q.size(0), H = q.size(1);
  const int T = q.size(2);
  at::Tensor dq = dq_out.has_value() ? *dq_out : at::empty_like(q);
  at::Tensor dk = dk_out.has_value() ? *dk_out : at::empty_like(k);
  at::Tensor dv = dv_out.has_value() ? *dv_out : at::empty_like(v);
  TORCH_CHECK(dq.strides() == dk.strides() && dq.strides() == dv.strides(),
              "dq/dk/dv must share strides");
  auto delta = at::empty({B * H * T}, q.options().dtype(at::kFloat));
  auto sq = strides3(q);
  auto so = strides3(o);
  auto sd = strides3(dq);
  check_hip(tdsa_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          o.data_ptr(), lsec.data_ptr<float>(), dout.data_ptr(),
                          dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                          delta.data_ptr<float>(), B, H, T, (float)scale,
                          sq.data(), so.data(), sd.data(), (float)dropout_p,
                          (unsigned long long)seed, cur_stream()),
            "attention_bwd");
  return {dq, dk, dv};
}

// ---- debug probes ---------------------------------------------------------
at::Tensor dbg_mfma(at::Tensor A, at::Tensor B, int64_t variant) {
  CHECK_IN(A); CHECK_IN(B);
  auto D = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  check_hip(tdsa_dbg_mfma(A.data_ptr(), B.data_ptr(), D.data_ptr<float>(),
                          (int)variant, cur_stream()),
            "dbg_mfma");
  return D;
}

at::Tensor dbg_mfma32(at::Tensor A, at::Tensor B) {
  CHECK_IN(A); CHECK_IN(B);
  auto D = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  check_hip(tdsa_dbg_mfma32(A.data_ptr(), B.data_ptr(), D.data_ptr<float>(),
                            cur_stream()),
            "dbg_mfma32");
  return D;
}

at::Tensor dbg_stage(at::Tensor in, int64_t transposed) {
  CHECK_IN(in);
  auto out = at::zeros_like(in);
  check_hip(tdsa_dbg_stage(in.data_ptr(), out.data_ptr(), (int)transposed,
                           cur_stream()),
            "dbg_stage");
  return out;
}

at::Tensor dbg_tr16(at::Tensor in) {
  CHECK_IN(in);
  auto out = at::zeros({2, 4, 2, 4, 64}, in.options().dtype(at::kFloat));
  check_hip(tdsa_dbg_tr16(in.data_ptr(), out.data_ptr<float>(), cur_stream()),
            "dbg_tr16");
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("dbg_mfma", &dbg_mfma);
  mod.def("dbg_mfma32", &dbg_mfma32);
  mod.def("dbg_stage", &dbg_stage);
  mod.def("dbg_tr16", &dbg_tr16);
  mod.def("layernorm_fwd", &layernorm_fwd, py::arg("x"), py::arg("w"),
          py::arg("b"), py::arg("eps"), py::arg("res") = py::none());
  mod.def("layernorm_bwd_dx", &layernorm_bwd_dx, py::arg("dy"), py::arg("x"),
          py::arg("w"), py::arg("mean"), py::arg("rstd"),
          py::arg("n_stripes"), py::arg("dh") = py::none());
  mod.def("layernorm_bwd_dwdb", &layernorm_bwd_dwdb);
  mod.def("gelu_fwd", &gelu_fwd);
  mod.def("gelu_bwd", &gelu_bwd);
  mod.def("column_sum", &column_sum);
  mod.def("embedding_fwd", &embedding_fwd);
  mod.def("embedding_bwd", &embedding_bwd);
  mod.def("cross_entropy_fwd", &cross_entropy_fwd);
  mod.def("cross_entropy_bwd", &cross_entropy_bwd, py::arg("logits"),
          py::arg("targets"), py::arg("lse"), py::arg("dloss"),
          py::arg("n_valid"), py::arg("ignore_index"),
          py::arg("out") = py::none());
  mod.def("gemm_tn", &gemm_tn);
  mod.def("adamw_step", &adamw_step);
  mod.def("adamw_step_multi", &adamw_step_multi);
  mod.def("sgd_step", &sgd_step);
  mod.def("sgd_step_multi", &sgd_step_multi);
  mod.def("attention_fwd", &attention_fwd, py::arg("q"), py::arg("k"),
          py::arg("v"), py::arg("scale"), py::arg("out") = py::none(),
          py::arg("dropout_p") = 0.0, py::arg("seed") = 0);
  mod.def("attention_bwd", &attention_bwd, py::arg("q"), py::arg("k"),
          py::arg("v"), py::arg("o"), py::arg("lse"), py::arg("dout"),
          py::arg("scale"), py::arg("dq") = py::none(),
          py::arg("dk") = py::none(), py::arg("dv") = py::none(),
          py::arg("dropout_p") = 0.0, py::arg("seed") = 0);
}
