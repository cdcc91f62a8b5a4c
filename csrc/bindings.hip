// Python bindings for the MI355X-native extension: RCCL comm core
// (csrc/comm/rccl_comm.hip) + CDNA4 HIP kernels (csrc/kernels/kernels.hip).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <cstdint>

namespace epl {
void register_comm(py::module& m);
}

extern "C" {
void epl_fused_adamw(float*, unsigned short*, const void*, bool, float*,
                     float*, int64_t, float, float, float, float, float, int,
                     float, hipStream_t);
void epl_lamb_phase1(const float*, const void*, bool, float*, float*, float*,
                     const int*, float*, float*, int64_t, float, float, float,
                     float, int, float, hipStream_t);
void epl_lamb_phase2(float*, unsigned short*, const float*, const int*,
                     const float*, int64_t, float, hipStream_t);
void epl_layer_norm_fwd(void*, const void*, const void*, void*, const void*,
                        const void*, float*, float*, int64_t, int64_t, float,
                        bool, hipStream_t);
void epl_layer_norm_bwd(void*, float*, float*, const void*, const void*,
                        const void*, const float*, const float*, float*,
                        float*, int64_t, int64_t, int64_t, bool,
                        hipStream_t);
void epl_bias_gelu_fwd(void*, const void*, const void*, int64_t, int64_t,
                       bool, hipStream_t);
void epl_bias_gelu_bwd(void*, float*, const void*, const void*, const void*,
                       float*, int64_t, int64_t, int64_t, bool,
                       hipStream_t);
void epl_ce_rowstats(const void*, const int64_t*, float*, float*, float*,
                     int64_t, int64_t, int64_t, int64_t, bool, hipStream_t);
void epl_ce_backward(void*, const void*, const int64_t*, const float*,
                     const float*, const float*, int64_t, int64_t, int64_t,
                     int64_t, float, bool, hipStream_t);
void epl_scale(void*, int64_t, float, bool, hipStream_t);
void epl_f32_to_bf16(unsigned short*, const float*, int64_t, hipStream_t);
void epl_bf16_to_f32(float*, const unsigned short*, int64_t, hipStream_t);
void epl_sqnorm(const void*, int64_t, float*, bool, hipStream_t);
void epl_colsum(void*, const void*, float*, int64_t, int64_t, int64_t,
                bool, hipStream_t);
void run_mfma_probe(const unsigned short*, const unsigned short*, float*,
                    hipStream_t);
void epl_moe_dispatch_fwd(void*, const void*, const int64_t*,
                          const int64_t*, const int64_t*, int64_t, int64_t,
                          int64_t, hipStream_t);
void epl_moe_dispatch_bwd(void*, const void*, const int64_t*,
                          const int64_t*, const int64_t*, int64_t, int64_t,
                          int64_t, int64_t, hipStream_t);
void epl_moe_combine_fwd(void*, const void*, const float*, const int64_t*,
                         const int64_t*, const int64_t*, int64_t, int64_t,
                         int64_t, int64_t, hipStream_t);
void epl_moe_combine_bwd(void*, float*, const void*, const void*,
                         const float*, const int64_t*, const int64_t*,
                         const int64_t*, int64_t, int64_t, int64_t,
                         hipStream_t);
void epl_attn_fwd(const void*, const void*, const void*, void*, float*,
                  int64_t, int64_t, float, bool, int64_t, int64_t,
                  const int64_t*,
                  const int64_t*, unsigned int*, int64_t,
                  unsigned long long, int, float, hipStream_t);
void epl_attn_bwd(const void*, const void*, const void*, const void*,
                  const void*, const float*, float*, void*, void*, void*,
                  int64_t, int64_t, float, bool, int64_t, const int64_t*,
                  const int64_t*, const int64_t*, const int64_t*, int,
                  const unsigned int*, int64_t, float, int64_t,
                  const float*, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

void check(const at::Tensor& t, at::ScalarType dtype, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dtype, name, " has dtype ", t.scalar_type(),
              ", expected ", dtype);
}

bool is_bf16(const at::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16 || t.scalar_type() == at::kFloat,
              "expected bf16 or fp32, got ", t.scalar_type());
  return t.scalar_type() == at::kBFloat16;
}

void fused_adamw(at::Tensor master, c10::optional<at::Tensor> param_bf16,
                 at::Tensor grad, at::Tensor m, at::Tensor v, double lr,
                 double beta1, double beta2, double eps, double weight_decay,
                 int64_t step, double inv_scale) {
  check(master, at::kFloat, "master");
  check(m, at::kFloat, "m");
  check(v, at::kFloat, "v");
  const bool gbf = is_bf16(grad);
  unsigned short* pb = nullptr;
  if (param_bf16.has_value()) {
    check(*param_bf16, at::kBFloat16, "param_bf16");
    TORCH_CHECK(param_bf16->numel() == master.numel());
    pb = reinterpret_cast<unsigned short*>(param_bf16->data_ptr());
  }
  const int64_t n = master.numel();
  TORCH_CHECK(grad.numel() == n && m.numel() == n && v.numel() == n,
              "arena size mismatch");
  epl_fused_adamw(master.data_ptr<float>(), pb, grad.data_ptr(), gbf,
                  m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
                  (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                  (int)step, (float)inv_scale, cur_stream());
}

void lamb_phase1(at::Tensor master, at::Tensor grad, at::Tensor m,
                 at::Tensor v, at::Tensor update, at::Tensor chunk_of,
                 at::Tensor wnorm_sq, at::Tensor unorm_sq, double beta1,
                 double beta2, double eps, double weight_decay, int64_t step,
                 double inv_scale) {
  check(master, at::kFloat, "master");
  check(update, at::kFloat, "update");
  check(chunk_of, at::kInt, "chunk_of");
  const bool gbf = is_bf16(grad);
  epl_lamb_phase1(master.data_ptr<float>(), grad.data_ptr(), gbf,
                  m.data_ptr<float>(), v.data_ptr<float>(),
                  update.data_ptr<float>(), chunk_of.data_ptr<int>(),
                  wnorm_sq.data_ptr<float>(), unorm_sq.data_ptr<float>(),
                  master.numel(), (float)beta1, (float)beta2, (float)eps,
                  (float)weight_decay, (int)step, (float)inv_scale,
                  cur_stream());
}

void lamb_phase2(at::Tensor master, c10::optional<at::Tensor> param_bf16,
                 at::Tensor update, at::Tensor chunk_of, at::Tensor ratio,
                 double lr) {
  check(master, at::kFloat, "master");
  unsigned short* pb = nullptr;
  if (param_bf16.has_value())
    pb = reinterpret_cast<unsigned short*>(param_bf16->data_ptr());
  epl_lamb_phase2(master.data_ptr<float>(), pb, update.data_ptr<float>(),
                  chunk_of.data_ptr<int>(), ratio.data_ptr<float>(),
                  master.numel(), (float)lr, cur_stream());
}

void layer_norm_fwd(at::Tensor out, at::Tensor x,
                    c10::optional<at::Tensor> res,
                    c10::optional<at::Tensor> sum_out, at::Tensor gamma,
                    at::Tensor beta, at::Tensor mean, at::Tensor rstd,
                    double eps) {
  const bool bf16 = is_bf16(x);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  TORCH_CHECK(!bf16 || cols % 8 == 0, "bf16 LayerNorm needs cols % 8 == 0");
  check(mean, at::kFloat, "mean");
  check(rstd, at::kFloat, "rstd");
  const void* res_p = nullptr;
  void* sum_p = nullptr;
  if (res.has_value()) {
    TORCH_CHECK(sum_out.has_value(),
                "fused residual LayerNorm needs sum_out");
    TORCH_CHECK(res->numel() == x.numel() && sum_out->numel() == x.numel());
    res_p = res->data_ptr();
    sum_p = sum_out->data_ptr();
  }
  epl_layer_norm_fwd(out.data_ptr(), x.data_ptr(), res_p, sum_p,
                     gamma.data_ptr(), beta.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), rows,
                     cols, (float)eps, bf16, cur_stream());
}

void layer_norm_bwd(at::Tensor dx, at::Tensor dgamma, at::Tensor dbeta,
                    at::Tensor dy, at::Tensor x, at::Tensor gamma,
                    at::Tensor mean, at::Tensor rstd) {
  const bool bf16 = is_bf16(x);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  TORCH_CHECK(!bf16 || cols % 8 == 0, "bf16 LayerNorm needs cols % 8 == 0");
  TORCH_CHECK((size_t)cols * 2 * sizeof(float) <= 128 * 1024,
              "LayerNorm backward supports cols <= 16384");
  check(dgamma, at::kFloat, "dgamma");
  check(dbeta, at::kFloat, "dbeta");
  float* dgp = nullptr;
  float* dbp = nullptr;
  at::Tensor ws;
  const bool fast = bf16 && cols % 8 == 0 && cols <= 2048;
  int64_t nparts = 0;
  if (fast) {
    const int64_t wave_rows = (rows + 3) / 4;
    nparts = (wave_rows < 1024 ? wave_rows : 1024) * 4;
    ws = at::empty({2, nparts, cols}, dgamma.options());
    dgp = ws.data_ptr<float>();
    dbp = dgp + nparts * cols;
  }
  epl_layer_norm_bwd(dx.data_ptr(), dgamma.data_ptr<float>(),
                     dbeta.data_ptr<float>(), dy.data_ptr(), x.data_ptr(),
                     gamma.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), dgp, dbp, nparts, rows, cols,
                     bf16, cur_stream());
}

void bias_gelu_fwd(at::Tensor out, at::Tensor x, at::Tensor bias) {
  const bool bf16 = is_bf16(x);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  TORCH_CHECK(!bf16 || cols % 8 == 0, "bf16 bias_gelu needs cols % 8 == 0");
  epl_bias_gelu_fwd(out.data_ptr(), x.data_ptr(), bias.data_ptr(), rows, cols,
                    bf16, cur_stream());
}

void bias_gelu_bwd(at::Tensor dx, at::Tensor dbias, at::Tensor dy,
                   at::Tensor x, at::Tensor bias) {
  const bool bf16 = is_bf16(x);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  TORCH_CHECK(!bf16 || cols % 8 == 0, "bf16 bias_gelu needs cols % 8 == 0");
  TORCH_CHECK((size_t)cols * sizeof(float) <= 128 * 1024,
              "bias_gelu backward supports cols <= 32768");
  check(dbias, at::kFloat, "dbias");
  float* dbp = nullptr;
  int64_t nwaves = 0;
  at::Tensor ws;
  if (bf16 && cols % 512 == 0) {
    const int64_t segs = cols / 512;
    int64_t per_seg = rows < 1024 ? rows : 1024;
    per_seg = (per_seg + 3) / 4 * 4;   // keep nwaves % (4*segs) clean
    nwaves = segs * per_seg;
    ws = at::empty({nwaves, 512}, dbias.options());
    dbp = ws.data_ptr<float>();
  }
  epl_bias_gelu_bwd(dx.data_ptr(), dbias.data_ptr<float>(), dy.data_ptr(),
                    x.data_ptr(), bias.data_ptr(), dbp, nwaves, rows, cols,
                    bf16, cur_stream());
}

void ce_rowstats(at::Tensor logits, at::Tensor targets, at::Tensor row_max,
                 at::Tensor row_sumexp, at::Tensor target_logit,
                 int64_t vocab_begin, int64_t ignore_index) {
  const bool bf16 = is_bf16(logits);
  const int64_t cols = logits.size(-1);
  const int64_t rows = logits.numel() / cols;
  check(targets, at::kLong, "targets");
  epl_ce_rowstats(logits.data_ptr(), targets.data_ptr<int64_t>(),
                  row_max.data_ptr<float>(), row_sumexp.data_ptr<float>(),
                  target_logit.data_ptr<float>(), rows, cols, vocab_begin,
                  ignore_index, bf16, cur_stream());
}

void ce_backward(at::Tensor dlogits, at::Tensor logits, at::Tensor targets,
                 at::Tensor gmax, at::Tensor gsumexp, at::Tensor dloss,
                 int64_t vocab_begin, int64_t ignore_index, double scale) {
  const bool bf16 = is_bf16(logits);
  const int64_t cols = logits.size(-1);
  const int64_t rows = logits.numel() / cols;
  epl_ce_backward(dlogits.data_ptr(), logits.data_ptr(),
                  targets.data_ptr<int64_t>(), gmax.data_ptr<float>(),
                  gsumexp.data_ptr<float>(), dloss.data_ptr<float>(), rows,
                  cols, vocab_begin, ignore_index, (float)scale, bf16,
                  cur_stream());
}

void scale_(at::Tensor t, double s) {
  const bool bf16 = is_bf16(t);
  epl_scale(t.data_ptr(), t.numel(), (float)s, bf16, cur_stream());
}

void f32_to_bf16(at::Tensor dst, at::Tensor src) {
  check(dst, at::kBFloat16, "dst");
  check(src, at::kFloat, "src");
  TORCH_CHECK(dst.numel() == src.numel());
  epl_f32_to_bf16(reinterpret_cast<unsigned short*>(dst.data_ptr()),
                  src.data_ptr<float>(), src.numel(), cur_stream());
}

void bf16_to_f32(at::Tensor dst, at::Tensor src) {
  check(dst, at::kFloat, "dst");
  check(src, at::kBFloat16, "src");
  TORCH_CHECK(dst.numel() == src.numel());
  epl_bf16_to_f32(dst.data_ptr<float>(),
                  reinterpret_cast<unsigned short*>(src.data_ptr()),
                  src.numel(), cur_stream());
}

static void attn_strides(const at::Tensor& t, int64_t* out3,
                         const char* name) {
  TORCH_CHECK(t.dim() == 4 && (t.size(3) == 64 || t.size(3) == 128), name,
              " must be [B,H,S,64|128]");
  TORCH_CHECK(t.stride(3) == 1, name, " last dim must be contiguous");
  TORCH_CHECK(t.stride(2) % 8 == 0 && t.stride(1) % 8 == 0,
              name, " row strides must be 16-byte aligned");
  out3[0] = t.stride(0);
  out3[1] = t.stride(1);
  out3[2] = t.stride(2);
}

// drop_mask: optional int32 [bh*seq*ceil(seq/32)] keep-bit tensor the
// kernel fills when dropout is on (drop_thresh in (0,256); the actual
// drop probability is drop_thresh/256).
void attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor out,
              at::Tensor lse, double scale, bool causal,
              c10::optional<at::Tensor> drop_mask, int64_t seed,
              int64_t drop_thresh, double inv_keep) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  int64_t in_s[3], o_s[3], tmp[3];
  attn_strides(q, in_s, "q");
  attn_strides(k, tmp, "k");
  TORCH_CHECK(tmp[0] == in_s[0] && tmp[1] == in_s[1] && tmp[2] == in_s[2],
              "q/k/v must share strides");
  attn_strides(v, tmp, "v");
  TORCH_CHECK(tmp[0] == in_s[0] && tmp[1] == in_s[1] && tmp[2] == in_s[2],
              "q/k/v must share strides");
  attn_strides(out, o_s, "out");
  check(lse, at::kFloat, "lse");
  const int64_t heads = q.size(1);
  const int64_t seq = q.size(2);
  const int64_t head_dim = q.size(3);
  const int64_t bh = q.size(0) * heads;
  unsigned int* mptr = nullptr;
  int64_t mask_w = (seq + 31) / 32;
  if (drop_mask.has_value()) {
    check(*drop_mask, at::kInt, "drop_mask");
    TORCH_CHECK(drop_mask->numel() == bh * seq * mask_w,
                "drop_mask size mismatch");
    TORCH_CHECK(drop_thresh > 0 && drop_thresh < 256, "drop_thresh range");
    mptr = reinterpret_cast<unsigned int*>(drop_mask->data_ptr());
  }
  epl_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
               lse.data_ptr<float>(), bh, seq, (float)scale, causal, heads,
               head_dim, in_s, o_s, mptr, mask_w,
               (unsigned long long)seed, (int)drop_thresh,
               (float)inv_keep, cur_stream());
}

void attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor out,
              at::Tensor dout, at::Tensor lse, at::Tensor delta_ws,
              at::Tensor dq, at::Tensor dk, at::Tensor dv, double scale,
              bool causal, bool split_dkdv,
              c10::optional<at::Tensor> drop_mask, double inv_keep,
              c10::optional<at::Tensor> dlse) {
  int64_t in_s[3], o_s[3], do_s[3], g_s[3], tmp[3];
  attn_strides(q, in_s, "q");
  attn_strides(out, o_s, "out");
  attn_strides(dout, do_s, "dout");
  attn_strides(dq, g_s, "dq");
  attn_strides(dk, tmp, "dk");
  TORCH_CHECK(tmp[0] == g_s[0] && tmp[1] == g_s[1] && tmp[2] == g_s[2],
              "dq/dk/dv must share strides");
  attn_strides(dv, tmp, "dv");
  TORCH_CHECK(tmp[0] == g_s[0] && tmp[1] == g_s[1] && tmp[2] == g_s[2],
              "dq/dk/dv must share strides");
  const int64_t heads = q.size(1);
  const int64_t seq = q.size(2);
  const int64_t bh = q.size(0) * heads;
  check(delta_ws, at::kFloat, "delta_ws");
  const unsigned int* mptr = nullptr;
  int64_t mask_w = (seq + 31) / 32;
  if (drop_mask.has_value()) {
    check(*drop_mask, at::kInt, "drop_mask");
    mptr = reinterpret_cast<const unsigned int*>(drop_mask->data_ptr());
  }
  const float* dlse_ptr = nullptr;
  if (dlse.has_value()) {
    check(*dlse, at::kFloat, "dlse");
    TORCH_CHECK(dlse->numel() == bh * seq, "dlse size mismatch");
    dlse_ptr = dlse->data_ptr<float>();
  }
  epl_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
               dout.data_ptr(), lse.data_ptr<float>(),
               delta_ws.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
               dv.data_ptr(), bh, seq, (float)scale, causal, heads, in_s,
               o_s, do_s, g_s, split_dkdv ? 1 : 0, mptr, mask_w,
               (float)inv_keep, q.size(3), dlse_ptr, cur_stream());
}

void mfma_probe(at::Tensor A, at::Tensor B, at::Tensor D) {
  check(A, at::kBFloat16, "A");
  check(B, at::kBFloat16, "B");
  check(D, at::kFloat, "D");
  run_mfma_probe(reinterpret_cast<const unsigned short*>(A.data_ptr()),
                 reinterpret_cast<const unsigned short*>(B.data_ptr()),
                 D.data_ptr<float>(), cur_stream());
}

static void check_idx(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
              t.scalar_type() == at::kLong, name,
              " must be a contiguous int64 device tensor");
}

void moe_dispatch_fwd(at::Tensor disp, at::Tensor x, at::Tensor fe,
                      at::Tensor pos, at::Tensor ft, int64_t cap) {
  check(disp, at::kBFloat16, "disp");
  check(x, at::kBFloat16, "x");
  check_idx(fe, "fe"); check_idx(pos, "pos"); check_idx(ft, "ft");
  const int64_t hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden % 8");
  epl_moe_dispatch_fwd(disp.data_ptr(), x.data_ptr(),
                       fe.data_ptr<int64_t>(), pos.data_ptr<int64_t>(),
                       ft.data_ptr<int64_t>(), fe.numel(), hidden, cap,
                       cur_stream());
}

void moe_dispatch_bwd(at::Tensor dx, at::Tensor ddisp, at::Tensor fe,
                      at::Tensor pos, at::Tensor inv, int64_t k,
                      int64_t cap) {
  check(dx, at::kBFloat16, "dx");
  check(ddisp, at::kBFloat16, "ddisp");
  check_idx(fe, "fe"); check_idx(pos, "pos"); check_idx(inv, "inv");
  const int64_t hidden = dx.size(-1);
  epl_moe_dispatch_bwd(dx.data_ptr(), ddisp.data_ptr(),
                       fe.data_ptr<int64_t>(), pos.data_ptr<int64_t>(),
                       inv.data_ptr<int64_t>(), dx.numel() / hidden, k,
                       hidden, cap, cur_stream());
}

void moe_combine_fwd(at::Tensor out, at::Tensor h, at::Tensor fw,
                     at::Tensor fe, at::Tensor pos, at::Tensor inv,
                     int64_t k, int64_t cap) {
  check(out, at::kBFloat16, "out");
  check(h, at::kBFloat16, "h");
  check(fw, at::kFloat, "fw");
  check_idx(fe, "fe"); check_idx(pos, "pos"); check_idx(inv, "inv");
  const int64_t hidden = out.size(-1);
  epl_moe_combine_fwd(out.data_ptr(), h.data_ptr(), fw.data_ptr<float>(),
                      fe.data_ptr<int64_t>(), pos.data_ptr<int64_t>(),
                      inv.data_ptr<int64_t>(), out.numel() / hidden, k,
                      hidden, cap, cur_stream());
}

void moe_combine_bwd(at::Tensor dh, at::Tensor dfw, at::Tensor dout,
                     at::Tensor h, at::Tensor fw, at::Tensor fe,
                     at::Tensor pos, at::Tensor ft, int64_t cap) {
  check(dh, at::kBFloat16, "dh");
  check(dout, at::kBFloat16, "dout");
  check(h, at::kBFloat16, "h");
  check(fw, at::kFloat, "fw");
  check(dfw, at::kFloat, "dfw");
  check_idx(fe, "fe"); check_idx(pos, "pos"); check_idx(ft, "ft");
  const int64_t hidden = dh.size(-1);
  epl_moe_combine_bwd(dh.data_ptr(), dfw.data_ptr<float>(),
                      dout.data_ptr(), h.data_ptr(), fw.data_ptr<float>(),
                      fe.data_ptr<int64_t>(), pos.data_ptr<int64_t>(),
                      ft.data_ptr<int64_t>(), fe.numel(), hidden, cap,
                      cur_stream());
}

void colsum(at::Tensor dy, at::Tensor db, at::Tensor partial) {
  const bool bf16 = is_bf16(dy);
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 2,
              "dy must be a contiguous 2-D device tensor");
  const int64_t rows = dy.size(0), cols = dy.size(1);
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(db.numel() == cols && db.scalar_type() == dy.scalar_type());
  const int64_t stripes = partial.numel() / cols;
  TORCH_CHECK(stripes >= 1 && partial.scalar_type() == at::kFloat);
  epl_colsum(db.data_ptr(), dy.data_ptr(), partial.data_ptr<float>(), rows,
             cols, stripes, bf16, cur_stream());
}

void sqnorm(at::Tensor t, at::Tensor out) {
  const bool bf16 = is_bf16(t);
  check(out, at::kFloat, "out");
  epl_sqnorm(t.data_ptr(), t.numel(), out.data_ptr<float>(), bf16,
             cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native parallel library: RCCL comm core + CDNA4 kernels";
  epl::register_comm(m);
  m.def("fused_adamw", &fused_adamw);
  m.def("colsum", &colsum);
  m.def("lamb_phase1", &lamb_phase1);
  m.def("lamb_phase2", &lamb_phase2);
  m.def("layer_norm_fwd", &layer_norm_fwd);
  m.def("layer_norm_bwd", &layer_norm_bwd);
  m.def("bias_gelu_fwd", &bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bias_gelu_bwd);
  m.def("ce_rowstats", &ce_rowstats);
  m.def("ce_backward", &ce_backward);
  m.def("scale_", &scale_);
  m.def("f32_to_bf16", &f32_to_bf16);
  m.def("bf16_to_f32", &bf16_to_f32);
  m.def("sqnorm", &sqnorm);
  m.def("mfma_probe", &mfma_probe);
  m.def("moe_dispatch_fwd", &moe_dispatch_fwd);
  m.def("moe_dispatch_bwd", &moe_dispatch_bwd);
  m.def("moe_combine_fwd", &moe_combine_fwd);
  m.def("moe_combine_bwd", &moe_combine_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
}
