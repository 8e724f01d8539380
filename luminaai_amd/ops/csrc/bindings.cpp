// PyTorch-ROCm bindings for the LuminaAI-AMD HIP kernels (gfx950).
// Host-side code only; kernels live in the *.hip translation units and are
// reached through extern "C" launchers taking hipStream_t.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <optional>
#include <tuple>

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_hip(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, what, " failed: ", hipGetErrorString(e));
}

bool is_bf16(const at::Tensor& t) { return t.scalar_type() == at::kBFloat16; }

void check_dtype(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16 || t.scalar_type() == at::kFloat,
              name, " must be bf16 or fp32, got ", t.scalar_type());
}

}  // namespace

// ---- launchers from the .hip TUs ----------------------------------------
extern "C" {
hipError_t lumina_attn_fwd(const void*, const void*, const void*, void*, void*, int, int, int, int, int, int, float, hipStream_t);
hipError_t lumina_attn_delta(const void*, const void*, void*, int64_t, int, int, int, hipStream_t);
hipError_t lumina_attn_bwd(const void*, const void*, const void*, const void*, const void*, const void*, void*, void*, void*, int, int, int, int, int, int, float, hipStream_t);
hipError_t lumina_rmsnorm_fwd_bf16(const void*, const void*, void*, float*, int64_t, int, float, hipStream_t);
hipError_t lumina_rmsnorm_fwd_f32(const void*, const void*, void*, float*, int64_t, int, float, hipStream_t);
hipError_t lumina_rmsnorm_bwd_bf16(const void*, const void*, const void*, const float*, void*, float*, int64_t, int, hipStream_t);
hipError_t lumina_rmsnorm_bwd_f32(const void*, const void*, const void*, const float*, void*, float*, int64_t, int, hipStream_t);
hipError_t lumina_rope_bf16(const void*, const void*, void*, void*, const float*, const float*, const int*, int64_t, int, int, int, int, int, int, hipStream_t);
hipError_t lumina_rope_f32(const void*, const void*, void*, void*, const float*, const float*, const int*, int64_t, int, int, int, int, int, int, hipStream_t);
hipError_t lumina_swiglu_fwd_bf16(const void*, const void*, void*, int64_t, int, int64_t, int64_t, hipStream_t);
hipError_t lumina_swiglu_fwd_f32(const void*, const void*, void*, int64_t, int, int64_t, int64_t, hipStream_t);
hipError_t lumina_swiglu_bwd_bf16(const void*, const void*, const void*, void*, void*, int64_t, int, int64_t, int64_t, hipStream_t);
hipError_t lumina_swiglu_bwd_f32(const void*, const void*, const void*, void*, void*, int64_t, int, int64_t, int64_t, hipStream_t);
hipError_t lumina_ce_fwd_bf16(const void*, const int32_t*, const float*, float*, float*, int64_t, int, int, hipStream_t);
hipError_t lumina_ce_fwd_f32(const void*, const int32_t*, const float*, float*, float*, int64_t, int, int, hipStream_t);
hipError_t lumina_ce_bwd_bf16(const void*, const int32_t*, const float*, const float*, const float*, const float*, void*, int64_t, int, int, hipStream_t);
hipError_t lumina_ce_bwd_f32(const void*, const int32_t*, const float*, const float*, const float*, const float*, void*, int64_t, int, int, hipStream_t);
hipError_t lumina_l2norm_sq_bf16(const void*, int64_t, float*, hipStream_t);
hipError_t lumina_l2norm_sq_f32(const void*, int64_t, float*, hipStream_t);
hipError_t lumina_adamw_step(float*, const void*, int, float*, float*, void*, int, int64_t, float, float, float, float, float, float, float, const float*, float, float, hipStream_t);
void launch_grouped_gemm_nt(const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, hipStream_t);
void launch_gg8p(const void*, const void*, void*, int, int, int, int, int, int64_t, int64_t, int64_t, int, hipStream_t);
hipError_t lumina_mx_quant_rows(const void*, void*, void*, int64_t, int, int, hipStream_t);
hipError_t lumina_mx_quant_cols(const void*, void*, void*, void*, int, int, int, int, hipStream_t);
hipError_t lumina_gg_mx_nt(const void*, const void*, const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, hipStream_t);
void launch_grouped_gemm_nt_v2(const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, int, hipStream_t);
void launch_grouped_gemm_nt_v3(const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, hipStream_t);
void launch_grouped_gemm_nt_v4(const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, hipStream_t);
hipError_t lumina_gemv(const void*, const void*, void*, int, int64_t, int, hipStream_t);
hipError_t lumina_dec_gemv(const void*, const void*, const void*, const void*, void*, int, int, float, int, hipStream_t);
hipError_t lumina_dec_rope_cache(const void*, void*, void*, void*, const float*, const float*, const int*, int, int, int, hipStream_t);
hipError_t lumina_dec_attn(const void*, const void*, const void*, void*, const int*, int, int, int, int, float, hipStream_t);
hipError_t lumina_dec_advance(int*, hipStream_t);
hipError_t lumina_gg8t_nt(const void*, const void*, void*, int, int, int, int, int64_t, int64_t, int64_t, int, hipStream_t);
hipError_t lumina_dec_rmsnorm(const void*, const void*, void*, int, float, hipStream_t);
hipError_t lumina_dec_topk(const void*, int*, float*, int, int, float, hipStream_t);
hipError_t lumina_dec_router(const void*, const void*, const void*, void*, int*, float*, int, int, int, float, float, hipStream_t);
hipError_t lumina_dec_gemv_moe(const void*, const void*, const void*, void*, const int*, const float*, int, int64_t, int, int, int, hipStream_t);
hipError_t lumina_moe_gather_rows(const void*, const int64_t*, const bool*, void*, int64_t, int, int, hipStream_t);
hipError_t lumina_moe_dispatch_bwd(const void*, const int64_t*, void*, int64_t, int, int, int64_t, int, hipStream_t);
hipError_t lumina_moe_combine_fwd(const void*, const float*, const int64_t*, void*, int64_t, int, int, int64_t, int, hipStream_t);
hipError_t lumina_moe_combine_bwd_y(const void*, const float*, const int64_t*, const int64_t*, const bool*, void*, int64_t, int64_t, int, int, hipStream_t);
hipError_t lumina_moe_combine_bwd_w(const void*, const void*, const int64_t*, float*, int64_t, int, int, int64_t, int, hipStream_t);
}

// ---- RMSNorm -------------------------------------------------------------
std::tuple<at::Tensor, at::Tensor> rmsnorm_fwd(const at::Tensor& x,
                                               const at::Tensor& w,
                                               double eps, bool save_inv) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  check_dtype(x, "x");
  TORCH_CHECK(x.scalar_type() == w.scalar_type(), "x/w dtype mismatch");
  const int H = (int)x.size(-1);
  const int64_t N = x.numel() / H;
  auto y = at::empty_like(x);
  auto inv = save_inv
      ? at::empty({N}, x.options().dtype(at::kFloat))
      : at::Tensor();
  float* invp = save_inv ? inv.data_ptr<float>() : nullptr;
  if (is_bf16(x))
    check_hip(lumina_rmsnorm_fwd_bf16(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                                      invp, N, H, (float)eps, cur_stream()),
              "rmsnorm_fwd_bf16");
  else
    check_hip(lumina_rmsnorm_fwd_f32(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                                     invp, N, H, (float)eps, cur_stream()),
              "rmsnorm_fwd_f32");
  return {y, inv};
}

std::tuple<at::Tensor, at::Tensor> rmsnorm_bwd(const at::Tensor& gy,
                                               const at::Tensor& x,
                                               const at::Tensor& w,
                                               const at::Tensor& inv) {
  TORCH_CHECK(gy.is_contiguous() && x.is_contiguous() && w.is_contiguous());
  const int H = (int)x.size(-1);
  const int64_t N = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  if (is_bf16(x))
    check_hip(lumina_rmsnorm_bwd_bf16(gy.data_ptr(), x.data_ptr(), w.data_ptr(),
                                      inv.data_ptr<float>(), dx.data_ptr(),
                                      dw.data_ptr<float>(), N, H, cur_stream()),
              "rmsnorm_bwd_bf16");
  else
    check_hip(lumina_rmsnorm_bwd_f32(gy.data_ptr(), x.data_ptr(), w.data_ptr(),
                                     inv.data_ptr<float>(), dx.data_ptr(),
                                     dw.data_ptr<float>(), N, H, cur_stream()),
              "rmsnorm_bwd_f32");
  return {dx, dw};
}

// ---- RoPE ----------------------------------------------------------------
// q: [B, S, Hq, D], k: [B, S, Hk, D] contiguous; cos/sin: [S_cache, D/2] fp32.
std::tuple<at::Tensor, at::Tensor> rope_fwd(const at::Tensor& q,
                                            const at::Tensor& k,
                                            const at::Tensor& cos_t,
                                            const at::Tensor& sin_t,
                                            const std::optional<at::Tensor>& pos,
                                            int64_t pos_offset, bool conj) {
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(cos_t.is_contiguous() && sin_t.is_contiguous());
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && sin_t.scalar_type() == at::kFloat);
  check_dtype(q, "q");
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4);
  const int64_t B = q.size(0);
  const int S = (int)q.size(1), Hq = (int)q.size(2), Hk = (int)k.size(2);
  const int D = (int)q.size(3);
  TORCH_CHECK(k.size(0) == B && k.size(1) == S && k.size(3) == D);
  TORCH_CHECK(cos_t.size(1) == D / 2);
  const int* posp = nullptr;
  if (pos.has_value()) {
    TORCH_CHECK(pos->scalar_type() == at::kInt && pos->is_contiguous());
    TORCH_CHECK(pos->numel() == B * S);
    posp = pos->data_ptr<int>();
  }
  auto oq = at::empty_like(q);
  auto ok = at::empty_like(k);
  if (is_bf16(q))
    check_hip(lumina_rope_bf16(q.data_ptr(), k.data_ptr(), oq.data_ptr(),
                               ok.data_ptr(), cos_t.data_ptr<float>(),
                               sin_t.data_ptr<float>(), posp, B, S, Hq, Hk, D,
                               (int)pos_offset, conj ? 1 : 0, cur_stream()),
              "rope_bf16");
  else
    check_hip(lumina_rope_f32(q.data_ptr(), k.data_ptr(), oq.data_ptr(),
                              ok.data_ptr(), cos_t.data_ptr<float>(),
                              sin_t.data_ptr<float>(), posp, B, S, Hq, Hk, D,
                              (int)pos_offset, conj ? 1 : 0, cur_stream()),
              "rope_f32");
  return {oq, ok};
}

// ---- SwiGLU ---------------------------------------------------------------
// g, u: [rows, I] views into the fused gate_up output (row stride may exceed I).
at::Tensor swiglu_fwd(const at::Tensor& g, const at::Tensor& u) {
  check_dtype(g, "gate");
  TORCH_CHECK(g.dim() == 2 && u.dim() == 2 && g.sizes() == u.sizes());
  TORCH_CHECK(g.stride(1) == 1 && u.stride(1) == 1, "inner dim must be contiguous");
  const int64_t rows = g.size(0);
  const int I = (int)g.size(1);
  auto y = at::empty({rows, I}, g.options());
  auto fn = is_bf16(g) ? lumina_swiglu_fwd_bf16 : lumina_swiglu_fwd_f32;
  check_hip(fn(g.data_ptr(), u.data_ptr(), y.data_ptr(), rows, I,
               g.stride(0), u.stride(0), cur_stream()), "swiglu_fwd");
  return y;
}

std::tuple<at::Tensor, at::Tensor> swiglu_bwd(const at::Tensor& dy,
                                              const at::Tensor& g,
                                              const at::Tensor& u) {
  TORCH_CHECK(dy.is_contiguous());
  TORCH_CHECK(g.stride(1) == 1 && u.stride(1) == 1);
  const int64_t rows = g.size(0);
  const int I = (int)g.size(1);
  // dg/du are allocated as one fused [rows, 2I] buffer when g/u share storage
  // layout, so the caller can feed the downstream GEMM one contiguous tensor.
  at::Tensor dg, du;
  bool fused = (g.stride(0) == 2 * I && u.stride(0) == 2 * I &&
                u.data_ptr() == (char*)g.data_ptr() + I * g.element_size());
  if (fused) {
    auto dgu = at::empty({rows, 2 * I}, g.options());
    dg = dgu.narrow(1, 0, I);
    du = dgu.narrow(1, I, I);
  } else {
    dg = at::empty({rows, I}, g.options());
    du = at::empty({rows, I}, u.options());
  }
  auto fn = is_bf16(g) ? lumina_swiglu_bwd_bf16 : lumina_swiglu_bwd_f32;
  check_hip(fn(dy.data_ptr(), g.data_ptr(), u.data_ptr(), dg.data_ptr(),
               du.data_ptr(), rows, I,
               fused ? 2 * I : I, fused ? 2 * I : I, cur_stream()),
            "swiglu_bwd");
  return {dg, du};
}

at::Tensor swiglu_bwd_fused(const at::Tensor& dy, const at::Tensor& gu) {
  // dy [rows, I], gu [rows, 2I] contiguous -> dgu [rows, 2I].  Taking the
  // FUSED projection output directly keeps autograd away from narrow()
  // slices: the two narrow-backwards cost a zero-fill + two slice copies
  // + a buffer add on [rows, 2I] (measured 6% of the b1 training step).
  TORCH_CHECK(dy.is_contiguous() && gu.is_contiguous());
  TORCH_CHECK(gu.dim() == 2 && dy.dim() == 2);
  const int64_t rows = gu.size(0);
  const int I = (int)(gu.size(1) / 2);
  TORCH_CHECK(dy.size(0) == rows && dy.size(1) == I);
  auto dgu = at::empty_like(gu);
  auto fn = is_bf16(gu) ? lumina_swiglu_bwd_bf16 : lumina_swiglu_bwd_f32;
  const char* g = (const char*)gu.data_ptr();
  char* dg = (char*)dgu.data_ptr();
  const int64_t ib = (int64_t)I * gu.element_size();
  check_hip(fn(dy.data_ptr(), g, g + ib, dg, dg + ib, rows, I,
               2 * (int64_t)I, 2 * (int64_t)I, cur_stream()),
            "swiglu_bwd_fused");
  return dgu;
}

// ---- fused CE ------------------------------------------------------------
std::tuple<at::Tensor, at::Tensor> ce_fwd(const at::Tensor& logits,
                                          const at::Tensor& labels,
                                          const std::optional<at::Tensor>& weights,
                                          int64_t ignore_index) {
  TORCH_CHECK(logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(labels.scalar_type() == at::kInt && labels.is_contiguous());
  check_dtype(logits, "logits");
  const int64_t N = logits.size(0);
  const int V = (int)logits.size(1);
  const float* wp = nullptr;
  if (weights.has_value()) {
    TORCH_CHECK(weights->scalar_type() == at::kFloat && weights->is_contiguous());
    wp = weights->data_ptr<float>();
  }
  auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
  auto stats = at::zeros({4}, logits.options().dtype(at::kFloat));
  auto fn = is_bf16(logits) ? lumina_ce_fwd_bf16 : lumina_ce_fwd_f32;
  check_hip(fn(logits.data_ptr(), labels.data_ptr<int>(), wp,
               lse.data_ptr<float>(), stats.data_ptr<float>(), N, V,
               (int)ignore_index, cur_stream()), "ce_fwd");
  return {lse, stats};
}

at::Tensor ce_bwd(const at::Tensor& logits, const at::Tensor& labels,
                  const std::optional<at::Tensor>& weights,
                  const at::Tensor& lse, const at::Tensor& stats,
                  const at::Tensor& gscale, int64_t ignore_index) {
  const int64_t N = logits.size(0);
  const int V = (int)logits.size(1);
  const float* wp = weights.has_value() ? weights->data_ptr<float>() : nullptr;
  auto dlogits = at::empty_like(logits);
  auto fn = is_bf16(logits) ? lumina_ce_bwd_bf16 : lumina_ce_bwd_f32;
  check_hip(fn(logits.data_ptr(), labels.data_ptr<int>(), wp,
               lse.data_ptr<float>(), stats.data_ptr<float>(),
               gscale.data_ptr<float>(), dlogits.data_ptr(), N, V,
               (int)ignore_index, cur_stream()), "ce_bwd");
  return dlogits;
}

// ---- optimizer -----------------------------------------------------------
at::Tensor l2norm_sq(const at::Tensor& x) {
  TORCH_CHECK(x.is_contiguous());
  check_dtype(x, "x");
  auto out = at::zeros({1}, x.options().dtype(at::kFloat));
  auto fn = is_bf16(x) ? lumina_l2norm_sq_bf16 : lumina_l2norm_sq_f32;
  check_hip(fn(x.data_ptr(), x.numel(), out.data_ptr<float>(), cur_stream()),
            "l2norm_sq");
  return out;
}

void adamw_step(at::Tensor& master, const at::Tensor& grad, at::Tensor& m,
                at::Tensor& v, const std::optional<at::Tensor>& w_out,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step, const std::optional<at::Tensor>& gnorm_sq,
                double max_norm, double grad_scale) {
  TORCH_CHECK(master.scalar_type() == at::kFloat && master.is_contiguous());
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  check_dtype(grad, "grad");
  const int64_t n = master.numel();
  TORCH_CHECK(grad.numel() == n && m.numel() == n && v.numel() == n);
  void* wp = nullptr;
  int w_bf16 = 1;
  if (w_out.has_value()) {
    TORCH_CHECK(w_out->numel() == n && w_out->is_contiguous());
    wp = w_out->data_ptr();
    w_bf16 = is_bf16(*w_out) ? 1 : 0;
  }
  const float* np = gnorm_sq.has_value() ? gnorm_sq->data_ptr<float>() : nullptr;
  const float bias1 = 1.0f / (1.0f - (float)std::pow(beta1, (double)step));
  const float bias2 = 1.0f / (1.0f - (float)std::pow(beta2, (double)step));
  check_hip(lumina_adamw_step(master.data_ptr<float>(), grad.data_ptr(),
                              is_bf16(grad) ? 1 : 0, m.data_ptr<float>(),
                              v.data_ptr<float>(), wp, w_bf16, n, (float)lr,
                              (float)beta1, (float)beta2, (float)eps, (float)wd,
                              bias1, bias2, np, (float)max_norm,
                              (float)grad_scale, cur_stream()),
            "adamw_step");
}

// ---- fused MoE dispatch/combine gathers ----------------------------------
at::Tensor moe_gather_rows(const at::Tensor& x, const at::Tensor& src_tok,
                           const at::Tensor& fill) {
  TORCH_CHECK(x.is_contiguous() && x.dim() == 2);
  const int64_t n_slots = src_tok.numel();
  auto buf = at::empty({n_slots, x.size(1)}, x.options());
  check_hip(lumina_moe_gather_rows(x.data_ptr(), src_tok.data_ptr<int64_t>(),
                                   fill.data_ptr<bool>(), buf.data_ptr(),
                                   n_slots, (int)x.size(1),
                                   is_bf16(x) ? 1 : 0, cur_stream()),
            "moe_gather_rows");
  return buf;
}

at::Tensor moe_dispatch_bwd(const at::Tensor& gbuf, const at::Tensor& slot_tm,
                            int64_t k) {
  TORCH_CHECK(gbuf.is_contiguous() && gbuf.dim() == 2);
  const int64_t n_tok = slot_tm.numel() / k;
  auto gx = at::empty({n_tok, gbuf.size(1)}, gbuf.options());
  check_hip(lumina_moe_dispatch_bwd(gbuf.data_ptr(),
                                    slot_tm.data_ptr<int64_t>(),
                                    gx.data_ptr(), n_tok, (int)k,
                                    (int)gbuf.size(1), gbuf.size(0),
                                    is_bf16(gbuf) ? 1 : 0, cur_stream()),
            "moe_dispatch_bwd");
  return gx;
}

at::Tensor moe_combine_fwd(const at::Tensor& y, const at::Tensor& w_tm,
                           const at::Tensor& slot_tm, int64_t k) {
  TORCH_CHECK(y.is_contiguous() && y.dim() == 2);
  TORCH_CHECK(w_tm.scalar_type() == at::kFloat);
  const int64_t n_tok = slot_tm.numel() / k;
  auto out = at::empty({n_tok, y.size(1)}, y.options());
  check_hip(lumina_moe_combine_fwd(y.data_ptr(), w_tm.data_ptr<float>(),
                                   slot_tm.data_ptr<int64_t>(),
                                   out.data_ptr(), n_tok, (int)k,
                                   (int)y.size(1), y.size(0),
                                   is_bf16(y) ? 1 : 0, cur_stream()),
            "moe_combine_fwd");
  return out;
}

at::Tensor moe_combine_bwd_y(const at::Tensor& gout, const at::Tensor& w_tm,
                             const at::Tensor& inv, const at::Tensor& src_tok,
                             const at::Tensor& fill) {
  TORCH_CHECK(gout.is_contiguous() && gout.dim() == 2);
  const int64_t n_slots = inv.numel();
  auto gy = at::empty({n_slots, gout.size(1)}, gout.options());
  check_hip(lumina_moe_combine_bwd_y(
                gout.data_ptr(), w_tm.data_ptr<float>(),
                inv.data_ptr<int64_t>(), src_tok.data_ptr<int64_t>(),
                fill.data_ptr<bool>(), gy.data_ptr(), n_slots,
                w_tm.numel(), (int)gout.size(1), is_bf16(gout) ? 1 : 0,
                cur_stream()),
            "moe_combine_bwd_y");
  return gy;
}

at::Tensor moe_combine_bwd_w(const at::Tensor& gout, const at::Tensor& y,
                             const at::Tensor& slot_tm, int64_t k) {
  TORCH_CHECK(gout.is_contiguous() && y.is_contiguous());
  const int64_t n_flat = slot_tm.numel();
  auto gw = at::empty({n_flat}, gout.options().dtype(at::kFloat));
  check_hip(lumina_moe_combine_bwd_w(
                gout.data_ptr(), y.data_ptr(), slot_tm.data_ptr<int64_t>(),
                gw.data_ptr<float>(), n_flat, (int)k, (int)gout.size(1),
                y.size(0), is_bf16(gout) ? 1 : 0, cur_stream()),
            "moe_combine_bwd_w");
  return gw;
}

// ---- grouped NT GEMM -----------------------------------------------------
at::Tensor grouped_gemm_nt(const at::Tensor& A, const at::Tensor& B) {
  // out[e] = A[e] @ B[e]^T : A [E,M,K], B [E,N,K] -> out [E,M,N]
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous(),
              "grouped_gemm_nt needs contiguous operands");
  TORCH_CHECK(is_bf16(A) && is_bf16(B), "grouped_gemm_nt is bf16-only");
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(0) == B.size(0)
              && A.size(2) == B.size(2), "shape mismatch");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  launch_grouped_gemm_nt(A.data_ptr(), B.data_ptr(), O.data_ptr(),
                         E, M, N, K, (int64_t)M * K, (int64_t)N * K,
                         (int64_t)M * N, cur_stream());
  check_hip(hipGetLastError(), "grouped_gemm_nt");
  return O;
}

at::Tensor grouped_gemm_nt_v2(const at::Tensor& A, const at::Tensor& B,
                              int64_t swz_mode) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(2) == B.size(2));
  TORCH_CHECK(A.size(2) % 64 == 0, "v2 kernel requires K % 64 == 0");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  launch_grouped_gemm_nt_v2(A.data_ptr(), B.data_ptr(), O.data_ptr(),
                            E, M, N, K, (int64_t)M * K, (int64_t)N * K,
                            (int64_t)M * N, (int)swz_mode, cur_stream());
  check_hip(hipGetLastError(), "grouped_gemm_nt_v2");
  return O;
}

at::Tensor grouped_gemm_nt_v3(const at::Tensor& A, const at::Tensor& B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(2) == B.size(2));
  TORCH_CHECK(A.size(2) % 64 == 0, "v3 kernel requires K % 64 == 0");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  launch_grouped_gemm_nt_v3(A.data_ptr(), B.data_ptr(), O.data_ptr(),
                            E, M, N, K, (int64_t)M * K, (int64_t)N * K,
                            (int64_t)M * N, cur_stream());
  check_hip(hipGetLastError(), "grouped_gemm_nt_v3");
  return O;
}

at::Tensor grouped_gemm_nt_v4(const at::Tensor& A, const at::Tensor& B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(2) == B.size(2));
  TORCH_CHECK(A.size(2) % 64 == 0, "v4 kernel requires K % 64 == 0");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  launch_grouped_gemm_nt_v4(A.data_ptr(), B.data_ptr(), O.data_ptr(),
                            E, M, N, K, (int64_t)M * K, (int64_t)N * K,
                            (int64_t)M * N, cur_stream());
  check_hip(hipGetLastError(), "grouped_gemm_nt_v4");
  return O;
}

// ---- causal GQA flash attention ------------------------------------------
// q,k,v: [B,H|HKV,S,DP] bf16 contiguous, DP % 32 == 0 (pad channels zero).
std::tuple<at::Tensor, at::Tensor> attn_fwd(const at::Tensor& q,
                                            const at::Tensor& k,
                                            const at::Tensor& v,
                                            double scale) {
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(is_bf16(q) && is_bf16(k) && is_bf16(v));
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  const int B = (int)q.size(0), H = (int)q.size(1);
  const int S = (int)q.size(2), DP = (int)q.size(3);
  const int HKV = (int)k.size(1);
  TORCH_CHECK(DP % 32 == 0 && (DP == 64 || DP == 128 || DP == 160),
              "unsupported padded head dim ", DP);
  TORCH_CHECK(H % HKV == 0 && k.size(2) == S && v.size(1) == HKV);
  const int SP = (S + 127) / 128 * 128;
  auto o = at::empty_like(q);
  auto lse2 = at::empty({B, H, SP}, q.options().dtype(at::kFloat));
  check_hip(lumina_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                            o.data_ptr(), lse2.data_ptr(), B, H, HKV, S, SP,
                            DP, (float)scale, cur_stream()),
            "attn_fwd");
  return {o, lse2};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> attn_bwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& do_, const at::Tensor& lse2,
    double scale) {
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous() &&
              o.is_contiguous() && do_.is_contiguous() && lse2.is_contiguous());
  TORCH_CHECK(is_bf16(do_) && lse2.scalar_type() == at::kFloat);
  const int B = (int)q.size(0), H = (int)q.size(1);
  const int S = (int)q.size(2), DP = (int)q.size(3);
  const int HKV = (int)k.size(1);
  const int SP = (S + 127) / 128 * 128;
  TORCH_CHECK(lse2.size(2) == SP);
  auto delta = at::empty({B, H, SP}, q.options().dtype(at::kFloat));
  check_hip(lumina_attn_delta(do_.data_ptr(), o.data_ptr(), delta.data_ptr(),
                              (int64_t)B * H * S, S, SP, DP, cur_stream()),
            "attn_delta");
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  check_hip(lumina_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                            do_.data_ptr(), lse2.data_ptr(), delta.data_ptr(),
                            dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                            B, H, HKV, S, SP, DP, (float)scale, cur_stream()),
            "attn_bwd");
  return {dq, dk, dv};
}

// deep-pipelined 256^2 grouped GEMM (gemm8p.hip)
// nt: out[e] = A[e] @ B[e]^T, A [E,M,K], B [E,N,K]; K % 32 == 0
// nn: out[e] = A[e] @ B[e],   A [E,M,K], B [E,Kb,N]; K % 32, Kb <= K
//     (A zero-padded along K up to a multiple of 32; B rows clamped)
at::Tensor gg8t_nt(const at::Tensor& A, const at::Tensor& B, int64_t pf2) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(2) == B.size(2));
  TORCH_CHECK(A.size(2) % 64 == 0, "gg8t requires K % 64 == 0");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  check_hip(lumina_gg8t_nt(A.data_ptr(), B.data_ptr(), O.data_ptr(), E, M,
                           N, K, (int64_t)M * K, (int64_t)N * K,
                           (int64_t)M * N, (int)pf2, cur_stream()),
            "gg8t_nt");
  return O;
}

at::Tensor gg8p_nt(const at::Tensor& A, const at::Tensor& B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && A.size(2) == B.size(2));
  TORCH_CHECK(A.size(2) % 32 == 0, "gg8p requires K % 32 == 0");
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), N = (int)B.size(1);
  auto O = at::empty({E, M, N}, A.options());
  launch_gg8p(A.data_ptr(), B.data_ptr(), O.data_ptr(), E, M, N, K, K,
              (int64_t)M * K, (int64_t)N * K, (int64_t)M * N, 0,
              cur_stream());
  check_hip(hipGetLastError(), "gg8p_nt");
  return O;
}

at::Tensor gg8p_nn(const at::Tensor& A, const at::Tensor& B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(is_bf16(A) && is_bf16(B));
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3);
  const int E = (int)A.size(0), M = (int)A.size(1);
  const int K = (int)A.size(2), Kb = (int)B.size(1), N = (int)B.size(2);
  TORCH_CHECK(K % 32 == 0, "gg8p requires (padded) K % 32 == 0");
  TORCH_CHECK(Kb <= K && K - Kb < 32, "K must be Kb padded up to 32");
  TORCH_CHECK(N % 8 == 0, "gg8p_nn requires N % 8 == 0");
  auto O = at::empty({E, M, N}, A.options());
  launch_gg8p(A.data_ptr(), B.data_ptr(), O.data_ptr(), E, M, N, K, Kb,
              (int64_t)M * K, (int64_t)Kb * N, (int64_t)M * N, 1,
              cur_stream());
  check_hip(hipGetLastError(), "gg8p_nn");
  return O;
}

// ---- MX-fp8 rowwise path (mxfp8.hip) -------------------------------------
std::tuple<at::Tensor, at::Tensor> mx_quant_rows(const at::Tensor& x,
                                                 int64_t kp) {
  TORCH_CHECK(x.is_contiguous() && is_bf16(x));
  const int K = (int)x.size(-1);
  const int Kp = (int)(kp > 0 ? kp : (K + 127) / 128 * 128);
  TORCH_CHECK(Kp % 128 == 0 && Kp >= K);
  const int64_t R = x.numel() / K;
  auto sizes = x.sizes().vec();
  sizes.back() = Kp;
  auto q = at::empty(sizes, x.options().dtype(at::kByte));
  auto ssz = x.sizes().vec();
  ssz.pop_back();
  auto s = at::empty(ssz, x.options().dtype(at::kByte));
  check_hip(lumina_mx_quant_rows(x.data_ptr(), q.data_ptr(), s.data_ptr(),
                                 R, K, Kp, cur_stream()),
            "mx_quant_rows");
  return {q, s};
}

std::tuple<at::Tensor, at::Tensor> mx_quant_cols(const at::Tensor& w,
                                                 int64_t kp) {
  // w [K, N] or batched [E, K, N] -> transposed quant [.., N, Kp]
  TORCH_CHECK(w.is_contiguous() && is_bf16(w));
  TORCH_CHECK(w.dim() == 2 || w.dim() == 3);
  const int E = w.dim() == 3 ? (int)w.size(0) : 1;
  const int K = (int)w.size(-2), N = (int)w.size(-1);
  const int Kp = (int)(kp > 0 ? kp : (K + 127) / 128 * 128);
  TORCH_CHECK(Kp % 128 == 0 && Kp >= K);
  auto q = w.dim() == 3
      ? at::empty({E, N, Kp}, w.options().dtype(at::kByte))
      : at::empty({N, Kp}, w.options().dtype(at::kByte));
  auto s = w.dim() == 3
      ? at::empty({E, N}, w.options().dtype(at::kByte))
      : at::empty({N}, w.options().dtype(at::kByte));
  auto ws = at::empty({(int64_t)E * N}, w.options().dtype(at::kFloat));
  check_hip(lumina_mx_quant_cols(w.data_ptr(), q.data_ptr(), s.data_ptr(),
                                 ws.data_ptr(), E, K, N, Kp, cur_stream()),
            "mx_quant_cols");
  return {q, s};
}

at::Tensor gg_mx_nt(const at::Tensor& Aq, const at::Tensor& As,
                    const at::Tensor& Bq, const at::Tensor& Bs) {
  TORCH_CHECK(Aq.is_contiguous() && Bq.is_contiguous());
  TORCH_CHECK(Aq.scalar_type() == at::kByte && Bq.scalar_type() == at::kByte);
  TORCH_CHECK(Aq.dim() == 3 && Bq.dim() == 3 && Aq.size(2) == Bq.size(2));
  TORCH_CHECK(Aq.size(2) % 128 == 0, "MX GEMM needs padded K % 128 == 0");
  const int E = (int)Aq.size(0), M = (int)Aq.size(1);
  const int K = (int)Aq.size(2), N = (int)Bq.size(1);
  TORCH_CHECK(As.numel() == (int64_t)E * M && Bs.numel() == (int64_t)E * N);
  auto O = at::empty({E, M, N},
                     Aq.options().dtype(at::kBFloat16));
  check_hip(lumina_gg_mx_nt(Aq.data_ptr(), Bq.data_ptr(), As.data_ptr(),
                            Bs.data_ptr(), O.data_ptr(), E, M, N, K,
                            (int64_t)M * K, (int64_t)N * K, (int64_t)M * N,
                            cur_stream()),
            "gg_mx_nt");
  return O;
}

// ---- fused batch-1 decode kernels (decode.hip) ---------------------------
void dec_gemv(const at::Tensor& W, const at::Tensor& x,
              const std::optional<at::Tensor>& wn,
              const std::optional<at::Tensor>& resid, at::Tensor& y,
              double eps, int64_t flags) {
  const int K = (int)x.numel();
  const int N = (int)y.numel();
  check_hip(lumina_dec_gemv(W.data_ptr(), x.data_ptr(),
                            wn ? wn->data_ptr() : nullptr,
                            resid ? resid->data_ptr() : nullptr,
                            y.data_ptr(), N, K, (float)eps, (int)flags,
                            cur_stream()),
            "dec_gemv");
}

void dec_rope_cache(const at::Tensor& qkv, at::Tensor& q_out, at::Tensor& kc,
                    at::Tensor& vc, const at::Tensor& cost,
                    const at::Tensor& sint, const at::Tensor& pos_dev,
                    int64_t H, int64_t HKV, int64_t D) {
  check_hip(lumina_dec_rope_cache(qkv.data_ptr(), q_out.data_ptr(),
                                  kc.data_ptr(), vc.data_ptr(),
                                  cost.data_ptr<float>(),
                                  sint.data_ptr<float>(),
                                  pos_dev.data_ptr<int>(), (int)H, (int)HKV,
                                  (int)D, cur_stream()),
            "dec_rope_cache");
}

void dec_attn(const at::Tensor& q, const at::Tensor& kc, const at::Tensor& vc,
              at::Tensor& out, const at::Tensor& pos_dev, int64_t H,
              int64_t HKV, int64_t D, double scale) {
  const int cap = (int)kc.size(0);
  check_hip(lumina_dec_attn(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                            out.data_ptr(), pos_dev.data_ptr<int>(), cap,
                            (int)H, (int)HKV, (int)D, (float)scale,
                            cur_stream()),
            "dec_attn");
}

void dec_advance(const at::Tensor& pos_dev) {
  check_hip(lumina_dec_advance(pos_dev.data_ptr<int>(), cur_stream()),
            "dec_advance");
}

void dec_rmsnorm(const at::Tensor& x, const at::Tensor& wn, at::Tensor& out,
                 double eps) {
  check_hip(lumina_dec_rmsnorm(x.data_ptr(), wn.data_ptr(), out.data_ptr(),
                               (int)x.numel(), (float)eps, cur_stream()),
            "dec_rmsnorm");
}

void dec_router(const at::Tensor& x, const at::Tensor& wn,
                const at::Tensor& Wg, at::Tensor& xhat, at::Tensor& eidx,
                at::Tensor& ew, int64_t k, double temp, double eps) {
  check_hip(lumina_dec_router(x.data_ptr(), wn.data_ptr(), Wg.data_ptr(),
                              xhat.data_ptr(), eidx.data_ptr<int>(),
                              ew.data_ptr<float>(), (int)x.numel(),
                              (int)Wg.size(0), (int)k, (float)temp,
                              (float)eps, cur_stream()),
            "dec_router");
}

void dec_topk(const at::Tensor& logits, at::Tensor& eidx, at::Tensor& ew,
              int64_t k, double temp) {
  check_hip(lumina_dec_topk(logits.data_ptr(), eidx.data_ptr<int>(),
                            ew.data_ptr<float>(), (int)logits.numel(),
                            (int)k, (float)temp, cur_stream()),
            "dec_topk");
}

void dec_gemv_moe(const at::Tensor& W, const at::Tensor& x,
                  const std::optional<at::Tensor>& resid, at::Tensor& y,
                  const at::Tensor& eidx, const at::Tensor& ew,
                  int64_t slot, int64_t flags) {
  // W: [E, N(or 2N), K] contiguous; rows per expert from y/x sizes
  const int K = (int)x.numel();
  const int N = (int)y.numel();
  const int64_t estride = W.size(1) * W.size(2);
  check_hip(lumina_dec_gemv_moe(W.data_ptr(), x.data_ptr(),
                                resid ? resid->data_ptr() : nullptr,
                                y.data_ptr(), eidx.data_ptr<int>(),
                                ew.data_ptr<float>(), (int)slot, estride,
                                N, K, (int)flags, cur_stream()),
            "dec_gemv_moe");
}

at::Tensor gemv(const at::Tensor& x, const at::Tensor& w) {
  // y[N] = w[N,K] @ x[K]
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.dim() == 1 && w.dim() == 2 && w.size(1) == x.numel());
  TORCH_CHECK(x.scalar_type() == w.scalar_type());
  auto y = at::empty({w.size(0)}, x.options());
  check_hip(lumina_gemv(w.data_ptr(), x.data_ptr(), y.data_ptr(),
                        (int)w.size(0), x.numel(), is_bf16(x) ? 1 : 0,
                        cur_stream()),
            "gemv");
  return y;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("attn_fwd", &attn_fwd,
          "causal GQA flash attention forward -> (O, LSE2) (gfx950)");
  mod.def("attn_bwd", &attn_bwd,
          "causal GQA flash attention backward -> (dQ, dK, dV) (gfx950)");
  mod.def("gemv", &gemv, "batch-1 decode GEMV y = W @ x (gfx950)");
  mod.def("dec_gemv", &dec_gemv,
          "fused decode GEMV (norm/residual/swiglu epilogues, nt loads)");
  mod.def("dec_rope_cache", &dec_rope_cache,
          "decode RoPE + KV-cache append at the device cursor");
  mod.def("dec_attn", &dec_attn, "single-token GQA attention over the cache");
  mod.def("dec_advance", &dec_advance, "advance the decode cursor");
  mod.def("swiglu_bwd_fused", &swiglu_bwd_fused,
          "swiglu backward into one fused [rows, 2I] grad buffer");
  mod.def("dec_rmsnorm", &dec_rmsnorm, "decode rmsnorm -> xhat buffer");
  mod.def("dec_router", &dec_router,
          "fused decode MoE router: rmsnorm + gate GEMV + top-k");
  mod.def("dec_topk", &dec_topk,
          "decode router: softmax/temp top-k, renormalized, on device");
  mod.def("dec_gemv_moe", &dec_gemv_moe,
          "expert-indirect decode GEMV (weight = W + eidx[slot]*estride)");
  mod.def("mx_quant_rows", &mx_quant_rows,
          "rowwise e8m0 fp8 quantization, K zero-padded (gfx950)");
  mod.def("mx_quant_cols", &mx_quant_cols,
          "columnwise e8m0 fp8 quantization + transpose (gfx950)");
  mod.def("gg_mx_nt", &gg_mx_nt,
          "grouped MX-fp8 GEMM A.B^T at the ~5PF fp8 MFMA rate (gfx950)");
  mod.def("gg8t_nt", &gg8t_nt,
          "8-phase-template grouped NT GEMM (256^2, BK64, 16x16x32)");
  mod.def("gg8p_nt", &gg8p_nt,
          "256^2 pipelined grouped GEMM, A.B^T (gfx950)");
  mod.def("gg8p_nn", &gg8p_nn,
          "256^2 pipelined grouped GEMM, A.B with K-major B (gfx950)");
  mod.def("grouped_gemm_nt_v4", &grouped_gemm_nt_v4,
          "2-buffer raw-barrier counted-vmcnt variant (K%64==0)");
  mod.def("grouped_gemm_nt_v3", &grouped_gemm_nt_v3,
          "3-buffer counted-vmcnt variant (K%64==0)");
  mod.def("grouped_gemm_nt", &grouped_gemm_nt,
          "grouped expert GEMM out[e]=A[e]@B[e]^T, bf16 MFMA (gfx950)");
  mod.def("grouped_gemm_nt_v2", &grouped_gemm_nt_v2,
          "32x32 MFMA variant (K%64==0), swizzle-mode knob");
  mod.def("moe_gather_rows", &moe_gather_rows, "MoE dispatch gather (gfx950)");
  mod.def("moe_dispatch_bwd", &moe_dispatch_bwd, "MoE dispatch backward");
  mod.def("moe_combine_fwd", &moe_combine_fwd, "MoE weighted combine");
  mod.def("moe_combine_bwd_y", &moe_combine_bwd_y, "MoE combine backward (y)");
  mod.def("moe_combine_bwd_w", &moe_combine_bwd_w, "MoE combine backward (w)");
  mod.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (gfx950)");
  mod.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (gfx950)");
  mod.def("rope_fwd", &rope_fwd, "RoPE q/k rotation (gfx950)");
  mod.def("swiglu_fwd", &swiglu_fwd, "SwiGLU forward (gfx950)");
  mod.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward (gfx950)");
  mod.def("ce_fwd", &ce_fwd, "fused CE+accuracy forward (gfx950)");
  mod.def("ce_bwd", &ce_bwd, "fused CE backward (gfx950)");
  mod.def("l2norm_sq", &l2norm_sq, "flat L2 norm squared (gfx950)");
  mod.def("adamw_step", &adamw_step, "fused clip+AdamW on flat buffers (gfx950)");
}
