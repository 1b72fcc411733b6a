// torch binding layer for paddle_amd._C (compiled host-side by g++;
// kernels live in kernels/*.hip compiled by hipcc for gfx950).
//
// Replaces the reference's pybind layer for our op set
// (paddle/fluid/pybind/eager_functions.cc + generated python_c) -- here
// the autograd glue lives in Python (torch.autograd.Function); this file
// only validates tensors and launches.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "kernels/api.h"

namespace pa {
void mfma_probe(const void* a, const void* bt, float* c, hipStream_t s);
void mfma_probe32(const void* a, const void* bt, float* c, hipStream_t s);
void mfma_probe_fp8mx(const void* a, const void* bt, float* c, int sa, int sb,
                      hipStream_t s);
}

namespace pa_lt {
std::tuple<torch::Tensor, torch::Tensor> fc1_gelu_fwd(
    const torch::Tensor& x, const torch::Tensor& w, const torch::Tensor& bias);
std::tuple<torch::Tensor, torch::Tensor> fc2_dgrad_dgelu(
    const torch::Tensor& dy, const torch::Tensor& w2, const torch::Tensor& z);
int64_t lt_epilogue_probe(int64_t m, int64_t n, int64_t k, int64_t epi);
}  // namespace pa_lt

namespace {

using torch::Tensor;

hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

int dt_of(const Tensor& t) {
  switch (t.scalar_type()) {
    case torch::kBFloat16: return pa::kBF16;
    case torch::kFloat: return pa::kF32;
    case torch::kHalf: return pa::kF16;
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
  return -1;
}

#define CHECK_IN(t) \
  TORCH_CHECK(t.is_cuda() && t.is_contiguous(), #t " must be contiguous GPU tensor")

// ---- norms ----------------------------------------------------------------
std::vector<Tensor> layer_norm_fwd(const Tensor& x, const Tensor& w,
                                   const c10::optional<Tensor>& b, double eps) {
  CHECK_IN(x); CHECK_IN(w);
  int64_t d = x.size(-1), n = x.numel() / d;
  TORCH_CHECK(d % 8 == 0, "layer_norm: D must be multiple of 8");
  auto y = torch::empty_like(x);
  auto mean = torch::empty({n}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({n}, x.options().dtype(torch::kFloat));
  pa::layer_norm_fwd(x.const_data_ptr(), w.const_data_ptr(),
                     b.has_value() ? b->const_data_ptr() : nullptr,
                     y.mutable_data_ptr(), mean.mutable_data_ptr<float>(),
                     rstd.mutable_data_ptr<float>(), n, d, (float)eps, dt_of(x),
                     cur_stream());
  return {y, mean, rstd};
}

std::vector<Tensor> layer_norm_bwd(const Tensor& dy, const Tensor& x,
                                   const Tensor& w, const Tensor& mean,
                                   const Tensor& rstd, bool has_bias) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(w);
  int64_t d = x.size(-1), n = x.numel() / d;
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({d}, x.options().dtype(torch::kFloat));
  auto db = torch::zeros({d}, x.options().dtype(torch::kFloat));
  pa::layer_norm_bwd_dx(dy.const_data_ptr(), x.const_data_ptr(), w.const_data_ptr(),
                        mean.const_data_ptr<float>(), rstd.const_data_ptr<float>(),
                        dx.mutable_data_ptr(), n, d, dt_of(x), cur_stream());
  auto ws = torch::empty({(int64_t)2 * pa::ln_dwdb_chunks(n, d) * d},
                         x.options().dtype(torch::kFloat));
  pa::layer_norm_bwd_dwdb(dy.const_data_ptr(), x.const_data_ptr(),
                          mean.const_data_ptr<float>(), rstd.const_data_ptr<float>(),
                          dw.mutable_data_ptr<float>(), db.mutable_data_ptr<float>(),
                          n, d, dt_of(x), cur_stream(), ws.mutable_data_ptr<float>());
  return {dx, dw.to(x.scalar_type()), db.to(x.scalar_type())};
}

std::vector<Tensor> rms_norm_fwd(const Tensor& x, const c10::optional<Tensor>& residual,
                                 const Tensor& w, double eps) {
  CHECK_IN(x); CHECK_IN(w);
  int64_t d = x.size(-1), n = x.numel() / d;
  TORCH_CHECK(d % 8 == 0, "rms_norm: D must be multiple of 8");
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({n}, x.options().dtype(torch::kFloat));
  Tensor res_out;
  const void* resp = nullptr;
  void* res_outp = nullptr;
  if (residual.has_value()) {
    CHECK_IN((*residual));
    res_out = torch::empty_like(x);
    resp = residual->const_data_ptr();
    res_outp = res_out.mutable_data_ptr();
  }
  pa::rms_norm_fwd(x.const_data_ptr(), resp, w.const_data_ptr(),
                   y.mutable_data_ptr(), res_outp, rstd.mutable_data_ptr<float>(),
                   n, d, (float)eps, dt_of(x), cur_stream());
  if (residual.has_value()) return {y, rstd, res_out};
  return {y, rstd};
}

std::vector<Tensor> rms_norm_bwd(const Tensor& dy, const Tensor& x, const Tensor& w,
                                 const Tensor& rstd) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(w);
  int64_t d = x.size(-1), n = x.numel() / d;
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({d}, x.options().dtype(torch::kFloat));
  pa::rms_norm_bwd_dx(dy.const_data_ptr(), x.const_data_ptr(), w.const_data_ptr(),
                      rstd.const_data_ptr<float>(), dx.mutable_data_ptr(), n, d,
                      dt_of(x), cur_stream());
  auto ws = torch::empty({(int64_t)2 * pa::ln_dwdb_chunks(n, d) * d},
                         x.options().dtype(torch::kFloat));
  pa::rms_norm_bwd_dw(dy.const_data_ptr(), x.const_data_ptr(),
                      rstd.const_data_ptr<float>(), dw.mutable_data_ptr<float>(),
                      n, d, dt_of(x), cur_stream(), ws.mutable_data_ptr<float>());
  return {dx, dw.to(x.scalar_type())};
}

// ---- cross entropy --------------------------------------------------------
std::vector<Tensor> softmax_ce_fwd(const Tensor& logits, const Tensor& labels,
                                   int64_t ignore_index) {
  CHECK_IN(logits); CHECK_IN(labels);
  int64_t v = logits.size(-1), n = logits.numel() / v;
  auto loss = torch::empty({n}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({n}, logits.options().dtype(torch::kFloat));
  pa::softmax_ce_fwd(logits.const_data_ptr(), labels.const_data_ptr<int64_t>(),
                     loss.mutable_data_ptr<float>(), lse.mutable_data_ptr<float>(),
                     n, v, ignore_index, dt_of(logits), cur_stream());
  return {loss, lse};
}

Tensor softmax_ce_bwd(const Tensor& dloss, const Tensor& logits,
                      const Tensor& labels, const Tensor& lse, int64_t ignore_index) {
  CHECK_IN(logits);
  int64_t v = logits.size(-1), n = logits.numel() / v;
  auto dlogits = torch::empty_like(logits);
  pa::softmax_ce_bwd(dloss.const_data_ptr<float>(), logits.const_data_ptr(),
                     labels.const_data_ptr<int64_t>(), lse.const_data_ptr<float>(),
                     dlogits.mutable_data_ptr(), n, v, ignore_index, dt_of(logits),
                     cur_stream());
  return dlogits;
}

// ---- elementwise ----------------------------------------------------------
Tensor bias_gelu_fwd(const Tensor& x, const c10::optional<Tensor>& bias) {
  CHECK_IN(x);
  int64_t d = x.size(-1), n = x.numel() / d;
  TORCH_CHECK(d % 8 == 0);
  auto y = torch::empty_like(x);
  pa::bias_gelu_fwd(x.const_data_ptr(), bias.has_value() ? bias->const_data_ptr() : nullptr,
                    y.mutable_data_ptr(), n, d, dt_of(x), cur_stream());
  return y;
}

Tensor bias_gelu_bwd(const Tensor& dy, const Tensor& x, const c10::optional<Tensor>& bias) {
  CHECK_IN(dy); CHECK_IN(x);
  int64_t d = x.size(-1), n = x.numel() / d;
  auto dx = torch::empty_like(x);
  pa::bias_gelu_bwd(dy.const_data_ptr(), x.const_data_ptr(),
                    bias.has_value() ? bias->const_data_ptr() : nullptr,
                    dx.mutable_data_ptr(), n, d, dt_of(x), cur_stream());
  return dx;
}

Tensor swiglu_fwd(const Tensor& x) {
  CHECK_IN(x);
  int64_t d2 = x.size(-1), n = x.numel() / d2, d = d2 / 2;
  TORCH_CHECK(d % 8 == 0);
  auto sizes = x.sizes().vec();
  sizes.back() = d;
  auto y = torch::empty(sizes, x.options());
  pa::swiglu_fwd(x.const_data_ptr(), y.mutable_data_ptr(), n, d, dt_of(x), cur_stream());
  return y;
}

Tensor swiglu_bwd(const Tensor& dy, const Tensor& x) {
  CHECK_IN(dy); CHECK_IN(x);
  int64_t d2 = x.size(-1), n = x.numel() / d2, d = d2 / 2;
  auto dx = torch::empty_like(x);
  pa::swiglu_bwd(dy.const_data_ptr(), x.const_data_ptr(), dx.mutable_data_ptr(),
                 n, d, dt_of(x), cur_stream());
  return dx;
}

Tensor rope_fwd(const Tensor& x, const Tensor& cos_t, const Tensor& sin_t,
                int64_t pos_offset, bool conj) {
  CHECK_IN(x); CHECK_IN(cos_t); CHECK_IN(sin_t);
  TORCH_CHECK(x.dim() == 4, "rope expects [B,S,H,D]");
  int64_t b = x.size(0), sl = x.size(1), h = x.size(2), dh = x.size(3);
  TORCH_CHECK((dh / 2) % 4 == 0, "rope: head_dim/2 must be multiple of 4");
  auto y = torch::empty_like(x);
  pa::rope_fwd(x.const_data_ptr(), cos_t.const_data_ptr<float>(),
               sin_t.const_data_ptr<float>(), y.mutable_data_ptr(), b, sl, h, dh,
               pos_offset, conj, dt_of(x), cur_stream());
  return y;
}

Tensor colsum(const Tensor& x) {
  CHECK_IN(x);
  int64_t d = x.size(-1), n = x.numel() / d;
  auto out = torch::zeros({d}, x.options().dtype(torch::kFloat));
  pa::colsum(x.const_data_ptr(), out.mutable_data_ptr<float>(), n, d, dt_of(x),
             cur_stream());
  return out;
}

Tensor weight_only_gemv(const Tensor& x, const Tensor& qweight,
                        const Tensor& scale, const c10::optional<Tensor>& bias) {
  CHECK_IN(x); CHECK_IN(qweight); CHECK_IN(scale);
  TORCH_CHECK(qweight.scalar_type() == torch::kChar, "qweight must be int8");
  int64_t k = x.size(-1), m = x.numel() / k, n = qweight.size(0);
  TORCH_CHECK(qweight.size(1) == k && k % 8 == 0);
  auto sizes = x.sizes().vec();
  sizes.back() = n;
  auto out = torch::empty(sizes, x.options());
  const void* bp = bias.has_value() ? bias->const_data_ptr() : nullptr;
  pa::weight_only_gemv(x.const_data_ptr(), qweight.const_data_ptr(),
                       scale.const_data_ptr<float>(), bp,
                       out.mutable_data_ptr(), m, n, k, dt_of(x), cur_stream());
  return out;
}

// ---- adamw ----------------------------------------------------------------
void adamw(Tensor& master, c10::optional<Tensor> param_out, const Tensor& grad,
           Tensor& m, Tensor& v, double lr, double beta1, double beta2,
           double eps, double wd, double beta1_pow, double beta2_pow,
           double grad_scale) {
  CHECK_IN(master); CHECK_IN(grad); CHECK_IN(m); CHECK_IN(v);
  int64_t numel = master.numel();
  bool bf16out = false;
  void* pout = nullptr;
  if (param_out.has_value()) {
    bf16out = param_out->scalar_type() == torch::kBFloat16;
    pout = param_out->mutable_data_ptr();
  }
  pa::adamw(master.mutable_data_ptr<float>(), pout, grad.const_data_ptr(),
            m.mutable_data_ptr<float>(), v.mutable_data_ptr<float>(), numel,
            (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd,
            (float)beta1_pow, (float)beta2_pow, (float)grad_scale,
            dt_of(grad), bf16out, cur_stream());
}

Tensor l2norm_sq(const Tensor& x) {
  CHECK_IN(x);
  auto out = torch::zeros({1}, x.options().dtype(torch::kFloat));
  pa::l2norm_sq(x.const_data_ptr(), out.mutable_data_ptr<float>(), x.numel(),
                dt_of(x), cur_stream());
  return out;
}

// ---- flash attention ------------------------------------------------------
// Tensors are logical [B, H, S, D] but may be strided VIEWS (e.g. of a packed
// [B, S, 3, H, D] qkv) as long as D is contiguous and rows are 16B-aligned.
static void fa_strides(const Tensor& t, int64_t* out) {
  TORCH_CHECK(t.stride(3) == 1, "flash_attn: head_dim must be contiguous");
  out[0] = t.stride(0);  // batch
  out[1] = t.stride(1);  // head
  out[2] = t.stride(2);  // seq
  TORCH_CHECK(out[2] % 8 == 0, "flash_attn: seq stride must keep 16B alignment");
}

static void fa_mask_strides(const Tensor& m, int64_t b, int64_t h, int64_t sq,
                            int64_t skv, int64_t* ms) {
  TORCH_CHECK(m.dim() == 4 && m.size(3) == skv && m.size(2) == sq &&
              (m.size(1) == 1 || m.size(1) == h) &&
              (m.size(0) == 1 || m.size(0) == b),
              "flash_attn mask must be [B|1, H|1, Sq, Skv]");
  TORCH_CHECK(m.stride(3) == 1 && m.scalar_type() == torch::kBFloat16,
              "flash_attn mask: bf16 with contiguous last dim");
  ms[0] = m.size(0) == 1 ? 0 : m.stride(0);
  ms[1] = m.size(1) == 1 ? 0 : m.stride(1);
  ms[2] = m.stride(2);
}

std::vector<Tensor> flash_attn_fwd(const Tensor& q, const Tensor& k, const Tensor& v,
                                   c10::optional<Tensor> o_out,
                                   double scale, bool causal,
                                   c10::optional<Tensor> mask,
                                   double pdrop, int64_t seed, int64_t offset) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16, "flash_attn: bf16 GPU only");
  TORCH_CHECK(q.dim() == 4, "flash_attn expects [B,H,S,D]");
  int64_t b = q.size(0), h = q.size(1), sq = q.size(2), d = q.size(3);
  int64_t hkv = k.size(1), skv = k.size(2);
  TORCH_CHECK(d == 128 || d == 64, "flash_attn: head_dim must be 64/128");
  TORCH_CHECK(h % hkv == 0);
  int64_t qs[3], ks[3], vs[3], os[3];
  fa_strides(q, qs); fa_strides(k, ks); fa_strides(v, vs);
  TORCH_CHECK(ks[0] == vs[0] && ks[1] == vs[1] && ks[2] == vs[2],
              "flash_attn: k/v must share strides");
  auto o = o_out.has_value() ? *o_out : torch::empty({b, h, sq, d}, q.options());
  fa_strides(o, os);
  auto lse = torch::empty({b, h, sq}, q.options().dtype(torch::kFloat));
  int64_t ms[3] = {0, 0, 0};
  const void* mp = nullptr;
  if (mask.has_value()) {
    fa_mask_strides(*mask, b, h, sq, skv, ms);
    mp = mask->const_data_ptr();
  }
  pa::flash_attn_fwd32(q.const_data_ptr(), k.const_data_ptr(), v.const_data_ptr(),
                       o.mutable_data_ptr(), lse.mutable_data_ptr<float>(), b, h,
                       hkv, sq, skv, d, (float)scale, causal, qs, ks, os,
                       mp, ms, (float)pdrop, (uint64_t)seed, (uint64_t)offset,
                       cur_stream());
  return {o, lse};
}

std::vector<Tensor> flash_attn_bwd(const Tensor& dout, const Tensor& q, const Tensor& k,
                                   const Tensor& v, const Tensor& o, const Tensor& lse,
                                   c10::optional<Tensor> dq_out,
                                   c10::optional<Tensor> dk_out,
                                   c10::optional<Tensor> dv_out,
                                   double scale, bool causal,
                                   c10::optional<Tensor> mask,
                                   double pdrop, int64_t seed, int64_t offset) {
  int64_t b = q.size(0), h = q.size(1), sq = q.size(2), d = q.size(3);
  int64_t hkv = k.size(1), skv = k.size(2);
  TORCH_CHECK(hkv == h, "flash_attn_bwd kernel requires hkv==h (expand KV upstream)");
  auto dq = dq_out.has_value() ? *dq_out : torch::empty({b, h, sq, d}, q.options());
  auto dk = dk_out.has_value() ? *dk_out : torch::empty({b, hkv, skv, d}, k.options());
  auto dv = dv_out.has_value() ? *dv_out : torch::empty({b, hkv, skv, d}, v.options());
  auto delta = torch::empty({b, h, sq}, q.options().dtype(torch::kFloat));
  int64_t qs[3], ks[3], vs[3], dos[3], os[3], dqs[3], dks[3], dvs[3];
  fa_strides(q, qs); fa_strides(k, ks); fa_strides(v, vs);
  fa_strides(dout, dos); fa_strides(o, os);
  fa_strides(dq, dqs); fa_strides(dk, dks); fa_strides(dv, dvs);
  TORCH_CHECK(ks[0] == vs[0] && ks[1] == vs[1] && ks[2] == vs[2]);
  TORCH_CHECK(dks[0] == dvs[0] && dks[1] == dvs[1] && dks[2] == dvs[2],
              "dk/dv must share strides");
  int64_t ms[3] = {0, 0, 0};
  const void* mp = nullptr;
  if (mask.has_value()) {
    fa_mask_strides(*mask, b, h, sq, skv, ms);
    mp = mask->const_data_ptr();
  }
  pa::flash_attn_bwd(dout.const_data_ptr(), q.const_data_ptr(), k.const_data_ptr(),
                     v.const_data_ptr(), o.const_data_ptr(), lse.const_data_ptr<float>(),
                     dq.mutable_data_ptr(), dk.mutable_data_ptr(), dv.mutable_data_ptr(),
                     delta.mutable_data_ptr<float>(), b, h, hkv, sq, skv, d,
                     (float)scale, causal, qs, ks, dos, os, dqs, dks,
                     mp, ms, (float)pdrop, (uint64_t)seed, (uint64_t)offset,
                     cur_stream());
  return {dq, dk, dv};
}

// host-side block map for the varlen kernels: (seq, offset) per block
static std::pair<Tensor, int64_t> make_bmap(const Tensor& cu_cpu, int64_t blk) {
  auto acc = cu_cpu.accessor<int, 1>();
  std::vector<int> map;
  for (int64_t i = 0; i + 1 < cu_cpu.size(0); ++i) {
    int len = acc[i + 1] - acc[i];
    for (int off = 0; off < len; off += (int)blk) {
      map.push_back((int)i);
      map.push_back(off);
    }
  }
  auto t = torch::from_blob(map.data(), {(int64_t)map.size()},
                            torch::dtype(torch::kInt32)).clone().cuda();
  return {t, (int64_t)map.size() / 2};
}

std::vector<Tensor> flash_attn_varlen_fwd(const Tensor& q, const Tensor& k,
                                          const Tensor& v, const Tensor& cu_q,
                                          const Tensor& cu_k, double scale,
                                          bool causal, double pdrop,
                                          int64_t seed, int64_t offset) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16 &&
              q.dim() == 3 && q.is_contiguous(), "varlen q: [total, H, D] bf16");
  TORCH_CHECK(cu_q.scalar_type() == torch::kInt32 && cu_k.scalar_type() == torch::kInt32);
  int64_t tq = q.size(0), h = q.size(1), d = q.size(2);
  int64_t tk = k.size(0), hkv = k.size(1);
  TORCH_CHECK(d == 128 || d == 64);
  auto cu_q_cpu = cu_q.cpu().contiguous();
  auto cu_k_cpu = cu_k.cpu().contiguous();
  auto cu_q_gpu = cu_q.is_cuda() ? cu_q.contiguous() : cu_q_cpu.cuda();
  auto cu_k_gpu = cu_k.is_cuda() ? cu_k.contiguous() : cu_k_cpu.cuda();
  auto [bmap, nblk] = make_bmap(cu_q_cpu, 128);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({h, tq}, q.options().dtype(torch::kFloat));
  pa::flash_attn_varlen_fwd32(q.const_data_ptr(), k.const_data_ptr(),
                              v.const_data_ptr(), o.mutable_data_ptr(),
                              lse.mutable_data_ptr<float>(), h, hkv, tq, tk, d,
                              (float)scale, causal, nblk,
                              cu_q_gpu.const_data_ptr<int>(),
                              cu_k_gpu.const_data_ptr<int>(),
                              bmap.const_data_ptr<int>(), (float)pdrop,
                              (uint64_t)seed, (uint64_t)offset, cur_stream());
  return {o, lse};
}

std::vector<Tensor> flash_attn_varlen_bwd(const Tensor& dout, const Tensor& q,
                                          const Tensor& k, const Tensor& v,
                                          const Tensor& o, const Tensor& lse,
                                          const Tensor& cu_q, const Tensor& cu_k,
                                          double scale, bool causal,
                                          double pdrop, int64_t seed,
                                          int64_t offset) {
  int64_t tq = q.size(0), h = q.size(1), d = q.size(2);
  int64_t tk = k.size(0);
  TORCH_CHECK(k.size(1) == h, "varlen bwd requires hkv == h");
  auto cu_q_cpu = cu_q.cpu().contiguous();
  auto cu_k_cpu = cu_k.cpu().contiguous();
  auto cu_q_gpu = cu_q.is_cuda() ? cu_q.contiguous() : cu_q_cpu.cuda();
  auto cu_k_gpu = cu_k.is_cuda() ? cu_k.contiguous() : cu_k_cpu.cuda();
  auto [qbmap, nqb] = make_bmap(cu_q_cpu, 128);
  auto [kvbmap, nkb] = make_bmap(cu_k_cpu, 128);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({h, tq}, q.options().dtype(torch::kFloat));
  pa::flash_attn_varlen_bwd(dout.contiguous().const_data_ptr(), q.const_data_ptr(),
                            k.const_data_ptr(), v.const_data_ptr(),
                            o.const_data_ptr(), lse.const_data_ptr<float>(),
                            dq.mutable_data_ptr(), dk.mutable_data_ptr(),
                            dv.mutable_data_ptr(), delta.mutable_data_ptr<float>(),
                            h, tq, tk, d, (float)scale, causal, nqb, nkb,
                            cu_q_gpu.const_data_ptr<int>(),
                            cu_k_gpu.const_data_ptr<int>(),
                            qbmap.const_data_ptr<int>(),
                            kvbmap.const_data_ptr<int>(), (float)pdrop,
                            (uint64_t)seed, (uint64_t)offset, cur_stream());
  return {dq, dk, dv};
}

Tensor fa_dropout_mask(int64_t b, int64_t h, int64_t sq, int64_t skv,
                       double p, int64_t seed, int64_t offset) {
  auto out = torch::empty({b, h, sq, skv},
                          torch::dtype(torch::kUInt8).device(torch::kCUDA));
  pa::fa_dropout_mask(out.mutable_data_ptr(), out.numel(), (float)p,
                      (uint64_t)seed, (uint64_t)offset, cur_stream());
  return out;
}

// ---- dropout_add ----------------------------------------------------------
std::vector<Tensor> dropout_add_fwd(const Tensor& x, const c10::optional<Tensor>& residual,
                                    double p, int64_t seed, int64_t offset) {
  CHECK_IN(x);
  auto y = torch::empty_like(x);
  Tensor mask;
  uint8_t* maskp = nullptr;
  if (p > 0) {
    mask = torch::empty(x.sizes(), x.options().dtype(torch::kUInt8));
    maskp = mask.mutable_data_ptr<uint8_t>();
  }
  pa::dropout_add_fwd(x.const_data_ptr(),
                      residual.has_value() ? residual->const_data_ptr() : nullptr,
                      y.mutable_data_ptr(), maskp, x.numel(), (float)p,
                      (uint64_t)seed, (uint64_t)offset, dt_of(x), cur_stream());
  if (p > 0) return {y, mask};
  return {y};
}

Tensor dropout_add_bwd(const Tensor& dy, const Tensor& mask, double p) {
  CHECK_IN(dy); CHECK_IN(mask);
  auto dx = torch::empty_like(dy);
  pa::dropout_add_bwd(dy.const_data_ptr(), mask.const_data_ptr<uint8_t>(),
                      dx.mutable_data_ptr(), dy.numel(), (float)p, dt_of(dy),
                      cur_stream());
  return dx;
}

// ---- embedding ------------------------------------------------------------
Tensor embedding_fwd(const Tensor& table, const Tensor& ids_in, int64_t padding_idx) {
  CHECK_IN(table);
  TORCH_CHECK(ids_in.is_cuda(), "ids must be on GPU");
  auto ids = ids_in.contiguous();
  int64_t vocab = table.size(0), d = table.size(1), n = ids.numel();
  TORCH_CHECK(d % 8 == 0);
  auto sizes = ids.sizes().vec();
  sizes.push_back(d);
  auto out = torch::empty(sizes, table.options());
  pa::embedding_fwd(table.const_data_ptr(), ids.const_data_ptr<int64_t>(),
                    out.mutable_data_ptr(), n, d, vocab, padding_idx,
                    dt_of(table), cur_stream());
  return out;
}

Tensor embedding_bwd(const Tensor& dout, const Tensor& ids_in, int64_t vocab,
                     int64_t padding_idx) {
  CHECK_IN(dout);
  TORCH_CHECK(ids_in.is_cuda(), "ids must be on GPU");
  auto ids = ids_in.contiguous();
  int64_t d = dout.size(-1), n = ids.numel();
  auto dtable = torch::zeros({vocab, d}, dout.options().dtype(torch::kFloat));
  pa::embedding_bwd(dout.const_data_ptr(), ids.const_data_ptr<int64_t>(),
                    dtable.mutable_data_ptr<float>(), n, d, vocab, padding_idx,
                    dt_of(dout), cur_stream());
  return dtable;
}

// ---- decode attention ------------------------------------------------------
Tensor decode_attention(const Tensor& q, const Tensor& kcache, const Tensor& vcache,
                        const Tensor& block_table, const Tensor& seq_lens,
                        double scale, int64_t blk_str, int64_t pos_str,
                        int64_t head_str, int64_t bs_override) {
  CHECK_IN(q); CHECK_IN(kcache); CHECK_IN(vcache);
  TORCH_CHECK(q.dim() == 3, "q: [B, H, D] (one decode step)");
  int64_t b = q.size(0), h = q.size(1), d = q.size(2);
  int64_t bs, hkv;
  if (blk_str == 0) {
    TORCH_CHECK(kcache.dim() == 4, "k_cache: [nblocks, block_size, HKV, D]");
    bs = kcache.size(1); hkv = kcache.size(2);
  } else {
    bs = bs_override; hkv = bs_override >= 0 ? h : h;  // dense: hkv == h
    hkv = kcache.size(1) == h ? h : kcache.size(1);
  }
  int64_t max_blocks = block_table.size(1);
  auto o = torch::empty_like(q);
  pa::decode_attention(q.const_data_ptr(), kcache.const_data_ptr(),
                       vcache.const_data_ptr(),
                       block_table.const_data_ptr<int>(),
                       seq_lens.const_data_ptr<int>(), o.mutable_data_ptr(),
                       b, h, hkv, bs, max_blocks, d, (float)scale,
                       blk_str, pos_str, head_str, cur_stream());
  return o;
}

// ---- hand-written GEMM ----------------------------------------------------
Tensor gemm_bf16(const Tensor& a, const Tensor& b, bool b_is_nt) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2);
  TORCH_CHECK(a.stride(1) == 1 && b.stride(1) == 1, "row-major inputs only");
  int64_t m = a.size(0), k = a.size(1);
  int64_t n = b_is_nt ? b.size(0) : b.size(1);
  TORCH_CHECK(b_is_nt ? b.size(1) == k : b.size(0) == k);
  TORCH_CHECK(k % 64 == 0 && k >= 128, "gemm_bf16: K must be multiple of 64");
  auto c = torch::empty({m, n}, a.options());
  pa::gemm_bf16(a.const_data_ptr(), b.const_data_ptr(), c.mutable_data_ptr(),
                m, n, k, a.stride(0), b.stride(0), c.stride(0), b_is_nt,
                cur_stream());
  return c;
}

// Full GEMM with layout + fused epilogue.  layout: 0=NT 1=NN 2=TN.
// epilogue: 0=none 1=bias 2=bias_gelu(aux out) 3=dgelu(aux in).
// Returns {C} or {C, aux} for bias_gelu.  c_in (optional): accumulate into
// it in-place (epilogue must be 0) -- the fused_linear_param_grad_add path.
std::vector<Tensor> gemm_bf16_ex(const Tensor& a, const Tensor& b,
                                 int64_t layout, int64_t epilogue,
                                 const c10::optional<Tensor>& bias,
                                 const c10::optional<Tensor>& aux_in,
                                 const c10::optional<Tensor>& c_in) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(b.is_cuda() && b.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2);
  TORCH_CHECK(a.stride(1) == 1 && b.stride(1) == 1, "row-major inputs only");
  int64_t m, n, k;
  if (layout == 0) {        // A[m][k] x Bt[n][k]
    m = a.size(0); k = a.size(1); n = b.size(0);
    TORCH_CHECK(b.size(1) == k, "gemm NT: inner dims mismatch");
  } else if (layout == 1) { // A[m][k] x B[k][n]
    m = a.size(0); k = a.size(1); n = b.size(1);
    TORCH_CHECK(b.size(0) == k, "gemm NN: inner dims mismatch");
  } else {                  // At[k][m] x B[k][n]
    k = a.size(0); m = a.size(1); n = b.size(1);
    TORCH_CHECK(b.size(0) == k, "gemm TN: inner dims mismatch");
  }
  TORCH_CHECK(k >= 64, "gemm_bf16_ex: K must be >= 64");
  bool accumulate = c_in.has_value();
  Tensor c;
  if (accumulate) {
    c = c_in.value();
    TORCH_CHECK(c.size(0) == m && c.size(1) == n && c.stride(1) == 1 &&
                c.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(epilogue == 0, "accumulate requires epilogue=none");
  } else {
    c = torch::empty({m, n}, a.options());
  }
  const void* bias_p = nullptr;
  if (epilogue == 1 || epilogue == 2) {
    TORCH_CHECK(bias.has_value() && bias->numel() == n &&
                bias->scalar_type() == torch::kBFloat16 && bias->is_contiguous());
    bias_p = bias->const_data_ptr();
  }
  Tensor aux;
  void* aux_p = nullptr;
  if (epilogue == 2) {
    aux = torch::empty({m, n}, a.options());
    aux_p = aux.mutable_data_ptr();
  } else if (epilogue == 3) {
    TORCH_CHECK(aux_in.has_value() && aux_in->size(0) == m &&
                aux_in->size(1) == n && aux_in->stride(1) == 1 &&
                aux_in->scalar_type() == torch::kBFloat16);
    aux_p = const_cast<void*>(aux_in->const_data_ptr());
  }
  pa::gemm_bf16_ex(a.const_data_ptr(), b.const_data_ptr(), c.mutable_data_ptr(),
                   bias_p, aux_p, m, n, k, a.stride(0), b.stride(0), c.stride(0),
                   (int)layout, (int)epilogue, accumulate, cur_stream());
  if (epilogue == 2) return {c, aux};
  return {c};
}

Tensor gemm_fp8_nt(const Tensor& a, const Tensor& bt, double scale_ab,
                   const c10::optional<Tensor>& bias) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && bt.dim() == 2);
  TORCH_CHECK(a.scalar_type() == torch::kFloat8_e4m3fn ||
              a.scalar_type() == torch::kUInt8, "gemm_fp8: e4m3/uint8 input");
  TORCH_CHECK(a.stride(1) == 1 && bt.stride(1) == 1, "row-major inputs");
  int64_t m = a.size(0), k = a.size(1), n = bt.size(0);
  TORCH_CHECK(bt.size(1) == k && k >= 128);
  auto c = torch::empty({m, n}, a.options().dtype(torch::kBFloat16));
  const void* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == torch::kBFloat16 && bias->numel() == n);
    bp = bias->const_data_ptr();
  }
  pa::gemm_fp8_nt(a.const_data_ptr(), bt.const_data_ptr(), c.mutable_data_ptr(),
                  bp, (float)scale_ab, m, n, k, a.stride(0), bt.stride(0),
                  c.stride(0), cur_stream());
  return c;
}

Tensor decode_gemm(const Tensor& x, const Tensor& w,
                   const c10::optional<Tensor>& bias) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && w.stride(1) == 1);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  int64_t m = x.size(0), k = x.size(1), n = w.size(1);
  TORCH_CHECK(w.size(0) == k && m <= 32);
  int64_t ntiles = (n + 1023) / 1024;
  int64_t ksplit = std::min<int64_t>(std::max<int64_t>(1, 768 / std::max<int64_t>(ntiles, 1)),
                                     std::max<int64_t>(1, k / 128));
  int64_t mt = m <= 8 ? 8 : (m <= 16 ? 16 : 32);
  auto ws = torch::empty({ksplit, mt, n}, x.options().dtype(torch::kFloat));
  auto y = torch::empty({m, n}, x.options());
  pa::decode_gemm(x.const_data_ptr(), w.const_data_ptr(),
                  bias.has_value() ? bias->const_data_ptr() : nullptr,
                  y.mutable_data_ptr(), ws.mutable_data_ptr<float>(), m, n, k,
                  w.stride(0), ksplit, cur_stream());
  return y;
}

Tensor decode_gemm_int8(const Tensor& x, const Tensor& qw,
                        const Tensor& scale,
                        const c10::optional<Tensor>& bias) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2 && qw.dim() == 2 && qw.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              qw.scalar_type() == torch::kChar &&
              scale.scalar_type() == torch::kFloat);
  int64_t m = x.size(0), k = x.size(1), n = qw.size(1);
  TORCH_CHECK(qw.size(0) == k && m <= 32 && n % 256 == 0 && k % 64 == 0);
  TORCH_CHECK(scale.numel() == n);
  int64_t nblk = n / 256;
  // fp32 partials cost ksplit*M*N*8 B of traffic -- cap harder (see the
  // bf16 binding's note; the reduce was 13% of serving kernel time)
  int64_t kmin = nblk <= 24 ? 256 : 512;
  int64_t ksplit = std::min<int64_t>(std::max<int64_t>(1, 1024 / nblk),
                                     std::max<int64_t>(1, k / kmin));
  int64_t mt = m <= 16 ? 16 : 32;
  auto ws = torch::empty({ksplit, mt, n}, x.options().dtype(torch::kFloat));
  auto y = torch::empty({m, n}, x.options());
  pa::decode_gemm_mfma(x.const_data_ptr(), qw.const_data_ptr(),
                       bias.has_value() ? bias->const_data_ptr() : nullptr,
                       y.mutable_data_ptr(), ws.mutable_data_ptr<float>(), m,
                       n, k, n, ksplit, cur_stream(),
                       scale.const_data_ptr<float>(), true);
  return y;
}

Tensor decode_gemm_mfma(const Tensor& x, const Tensor& w,
                        const c10::optional<Tensor>& bias) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && w.stride(1) == 1);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  int64_t m = x.size(0), k = x.size(1), n = w.size(1);
  TORCH_CHECK(w.size(0) == k && m <= 32 && n % 256 == 0 && k % 64 == 0);
  // target ~1024 workgroups, but keep kchunk >= 256 (4 pipelined LDS
  // tiles per block; kchunk == 64 degenerates to an unpipelined tile)
  int64_t nblk = n / 256;
  // small-N shapes need a denser grid: allow kchunk 128 to reach 2 wg/CU.
  // Cap ksplit harder than the grid target alone suggests: the fp32
  // partial buffer costs ksplit*M*N*8 B of write+read, which at int8
  // weights rivals the weight stream itself (the reduce kernel measured
  // 13% of serving kernel time at the 1024-wg target).
  int64_t kmin = nblk <= 24 ? 256 : 512;
  int64_t ksplit = std::min<int64_t>(std::max<int64_t>(1, 1024 / nblk),
                                     std::max<int64_t>(1, k / kmin));
  int64_t mt = m <= 16 ? 16 : 32;
  auto ws = torch::empty({ksplit, mt, n}, x.options().dtype(torch::kFloat));
  auto y = torch::empty({m, n}, x.options());
  pa::decode_gemm_mfma(x.const_data_ptr(), w.const_data_ptr(),
                       bias.has_value() ? bias->const_data_ptr() : nullptr,
                       y.mutable_data_ptr(), ws.mutable_data_ptr<float>(), m,
                       n, k, w.stride(0), ksplit, cur_stream());
  return y;
}

// ---- MoE routing ----------------------------------------------------------
std::vector<Tensor> moe_gate_topk(const Tensor& logits, int64_t k) {
  CHECK_IN(logits);
  TORCH_CHECK(logits.scalar_type() == torch::kFloat && logits.dim() == 2);
  int64_t t = logits.size(0), e = logits.size(1);
  TORCH_CHECK(e <= 64 && k <= 4);
  auto topv = torch::empty({t, k}, logits.options());
  auto topi = torch::empty({t, k}, logits.options().dtype(torch::kInt32));
  auto me = torch::zeros({e}, logits.options());
  auto ce = torch::zeros({e}, logits.options());
  pa::moe_gate_topk(logits.const_data_ptr<float>(), topv.mutable_data_ptr<float>(),
                    topi.mutable_data_ptr<int>(), me.mutable_data_ptr<float>(),
                    ce.mutable_data_ptr<float>(), t, e, k, cur_stream());
  return {topv, topi, me, ce};
}

std::vector<Tensor> moe_assign_slots(const Tensor& topi, int64_t num_experts,
                                     int64_t cap) {
  CHECK_IN(topi);
  TORCH_CHECK(topi.scalar_type() == torch::kInt32 && topi.dim() == 2);
  int64_t t = topi.size(0), k = topi.size(1);
  auto slot = torch::full({t, k}, -1, topi.options());
  auto counts = torch::zeros({num_experts}, topi.options());
  pa::moe_assign_slots(topi.const_data_ptr<int>(), slot.mutable_data_ptr<int>(),
                       counts.mutable_data_ptr<int>(), t, num_experts, k, cap,
                       cur_stream());
  return {slot, counts};
}

Tensor gemm_fp8_nt_batched(const Tensor& a, const Tensor& bt, double scale_ab,
                           const c10::optional<Tensor>& bias) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 3 && bt.dim() == 3);
  TORCH_CHECK(a.is_contiguous() && bt.is_contiguous());
  int64_t e = a.size(0), m = a.size(1), k = a.size(2), n = bt.size(1);
  TORCH_CHECK(bt.size(0) == e && bt.size(2) == k && k >= 128);
  const void* bp = nullptr;
  int64_t bias_bs = 0;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_contiguous() && bias->scalar_type() == torch::kBFloat16);
    TORCH_CHECK(bias->dim() == 2 && bias->size(0) == e && bias->size(1) == n);
    bp = bias->const_data_ptr();
    bias_bs = n;
  }
  auto c = torch::empty({e, m, n}, a.options().dtype(torch::kBFloat16));
  pa::gemm_fp8_nt_batched(a.const_data_ptr(), bt.const_data_ptr(),
                          c.mutable_data_ptr(), bp, (float)scale_ab, e,
                          m, n, k, k, k, n, m * k, n * k, m * n, cur_stream(),
                          bias_bs);
  return c;
}

Tensor amax_abs(const Tensor& x) {
  CHECK_IN(x);
  auto out = torch::zeros({}, x.options().dtype(torch::kFloat));
  pa::amax_abs(x.const_data_ptr(), out.mutable_data_ptr<float>(), x.numel(),
               dt_of(x), cur_stream());
  return out;
}

Tensor gemm_bf16_8p_t(const Tensor& a, const Tensor& bt) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && bt.dim() == 2);
  TORCH_CHECK(a.is_contiguous() && bt.is_contiguous());
  int64_t m = a.size(0), k = a.size(1), n = bt.size(0);
  TORCH_CHECK(bt.size(1) == k && m % 256 == 0 && n % 256 == 0 && k % 32 == 0);
  auto c = torch::empty({m, n}, a.options());
  pa::gemm_bf16_8p(a.const_data_ptr(), bt.const_data_ptr(),
                   c.mutable_data_ptr(), m, n, k, k, k, n, cur_stream());
  return c;
}

Tensor gemm_bf16_nt_batched_t(const Tensor& a, const Tensor& bt) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 3 && bt.dim() == 3);
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
              bt.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.is_contiguous() && bt.is_contiguous());
  int64_t e = a.size(0), m = a.size(1), k = a.size(2), n = bt.size(1);
  TORCH_CHECK(bt.size(0) == e && bt.size(2) == k && k % 8 == 0 && n % 8 == 0);
  auto c = torch::empty({e, m, n}, a.options());
  pa::gemm_bf16_nt_batched(a.const_data_ptr(), bt.const_data_ptr(),
                           c.mutable_data_ptr(), e, m, n, k, k, k, n,
                           m * k, n * k, m * n, cur_stream());
  return c;
}

Tensor quant_fp8(const Tensor& x, double scale) {
  CHECK_IN(x);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.numel() % 8 == 0);
  auto out = torch::empty(x.sizes(), x.options().dtype(torch::kFloat8_e4m3fn));
  pa::quant_fp8(x.const_data_ptr(), out.mutable_data_ptr(), (float)scale,
                x.numel(), cur_stream());
  return out;
}

// ---- probe ----------------------------------------------------------------
Tensor mfma_probe_fp8mx(const Tensor& a, const Tensor& bt, int64_t sa, int64_t sb) {
  CHECK_IN(a); CHECK_IN(bt);
  TORCH_CHECK(a.scalar_type() == torch::kUInt8 && a.numel() == 16 * 128);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  pa::mfma_probe_fp8mx(a.const_data_ptr(), bt.const_data_ptr(),
                       c.mutable_data_ptr<float>(), (int)sa, (int)sb,
                       cur_stream());
  return c;
}

Tensor mfma_probe(const Tensor& a, const Tensor& bt) {
  CHECK_IN(a); CHECK_IN(bt);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  pa::mfma_probe(a.const_data_ptr(), bt.const_data_ptr(),
                 c.mutable_data_ptr<float>(), cur_stream());
  return c;
}

Tensor mfma_probe32(const Tensor& a, const Tensor& bt) {
  CHECK_IN(a); CHECK_IN(bt);
  auto c = torch::empty({32, 32}, a.options().dtype(torch::kFloat));
  pa::mfma_probe32(a.const_data_ptr(), bt.const_data_ptr(),
                   c.mutable_data_ptr<float>(), cur_stream());
  return c;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layer_norm_fwd", &layer_norm_fwd);
  m.def("layer_norm_bwd", &layer_norm_bwd);
  m.def("rms_norm_fwd", &rms_norm_fwd);
  m.def("rms_norm_bwd", &rms_norm_bwd);
  m.def("softmax_ce_fwd", &softmax_ce_fwd);
  m.def("softmax_ce_bwd", &softmax_ce_bwd);
  m.def("bias_gelu_fwd", &bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bias_gelu_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("colsum", &colsum);
  m.def("adamw", &adamw);
  m.def("fc1_gelu_fwd", &pa_lt::fc1_gelu_fwd);
  m.def("fc2_dgrad_dgelu", &pa_lt::fc2_dgrad_dgelu);
  m.def("lt_epilogue_probe", &pa_lt::lt_epilogue_probe);
  m.def("weight_only_gemv", &weight_only_gemv);
  m.def("l2norm_sq", &l2norm_sq);
  m.def("flash_attn_fwd", &flash_attn_fwd, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o") = c10::nullopt, py::arg("scale"),
        py::arg("causal"), py::arg("mask") = c10::nullopt,
        py::arg("pdrop") = 0.0, py::arg("seed") = 0, py::arg("offset") = 0);
  m.def("fa_dropout_mask", &fa_dropout_mask);
  m.def("mfma_probe_fp8mx", &mfma_probe_fp8mx);
  m.def("decode_gemm", &decode_gemm, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt);
  m.def("decode_gemm_mfma", &decode_gemm_mfma, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt);
  m.def("decode_gemm_int8", &decode_gemm_int8, py::arg("x"), py::arg("qw"),
        py::arg("scale"), py::arg("bias") = c10::nullopt);
  m.def("moe_gate_topk", &moe_gate_topk);
  m.def("moe_assign_slots", &moe_assign_slots);
  m.def("gemm_fp8_nt", &gemm_fp8_nt, py::arg("a"), py::arg("bt"),
        py::arg("scale_ab") = 1.0, py::arg("bias") = c10::nullopt);
  m.def("gemm_fp8_nt_batched", &gemm_fp8_nt_batched, py::arg("a"),
        py::arg("bt"), py::arg("scale_ab"), py::arg("bias") = c10::nullopt);
  m.def("amax_abs", &amax_abs);
  m.def("quant_fp8", &quant_fp8);
  m.def("flash_attn_varlen_fwd", &flash_attn_varlen_fwd, py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("cu_q"), py::arg("cu_k"),
        py::arg("scale"), py::arg("causal"), py::arg("pdrop") = 0.0,
        py::arg("seed") = 0, py::arg("offset") = 0);
  m.def("flash_attn_varlen_bwd", &flash_attn_varlen_bwd);
  m.def("flash_attn_bwd", &flash_attn_bwd, py::arg("dout"), py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("o"), py::arg("lse"),
        py::arg("dq") = c10::nullopt, py::arg("dk") = c10::nullopt,
        py::arg("dv") = c10::nullopt, py::arg("scale"), py::arg("causal"),
        py::arg("mask") = c10::nullopt, py::arg("pdrop") = 0.0,
        py::arg("seed") = 0, py::arg("offset") = 0);
  m.def("dropout_add_fwd", &dropout_add_fwd);
  m.def("dropout_add_bwd", &dropout_add_bwd);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("gemm_bf16_nt_batched", &gemm_bf16_nt_batched_t);
  m.def("gemm_bf16_8p", &gemm_bf16_8p_t);
  m.def("gemm_bf16_ex", &gemm_bf16_ex, py::arg("a"), py::arg("b"),
        py::arg("layout"), py::arg("epilogue") = 0,
        py::arg("bias") = c10::nullopt, py::arg("aux") = c10::nullopt,
        py::arg("c_in") = c10::nullopt);
  m.def("decode_attention", &decode_attention, py::arg("q"), py::arg("kcache"),
        py::arg("vcache"), py::arg("block_table"), py::arg("seq_lens"),
        py::arg("scale"), py::arg("blk_str") = 0, py::arg("pos_str") = 0,
        py::arg("head_str") = 0, py::arg("bs_override") = 0);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma_probe32", &mfma_probe32);
  m.attr("compiled_arch") = "gfx950";
}
