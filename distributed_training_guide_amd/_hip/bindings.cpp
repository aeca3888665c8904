// PyTorch bindings for the gfx950 HIP kernels (compiled with hipcc,
// linked against libtorch; no hipify, no CUDA compatibility layer —
// c10::hip is the native ROCm stream API).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
void rmsnorm_fwd_launch(const void*, const void*, void*, void*, int64_t, int,
                        float, hipStream_t);
void rmsnorm_bwd_launch(const void*, const void*, const void*, const void*,
                        void*, float*, void*, int, int64_t, int, hipStream_t);
void qkv_rope_fwd_launch(const void*, void*, void*, void*, const float*,
                         const float*, const int*, int64_t, int, int, int,
                         int, hipStream_t);
void qkv_rope_bwd_launch(const void*, const void*, const void*, void*,
                         const float*, const float*, const int*, int64_t,
                         int, int, int, int, hipStream_t);
void add_rmsnorm_fwd_launch(const void*, const void*, const void*, void*,
                            void*, void*, int64_t, int, float, hipStream_t);
void add_rmsnorm_bwd_launch(const void*, const void*, const void*,
                            const void*, const void*, void*, float*, void*,
                            int, int64_t, int, hipStream_t);
void rmsnorm_dw_reduce_launch(const float*, void*, int, int, hipStream_t);
void rope_launch(const void*, void*, const float*, const float*, const int*,
                 int64_t, int, int, int, int, hipStream_t);
void silu_mul_fwd_launch(const void*, void*, int64_t, int, hipStream_t);
void silu_mul_bwd_launch(const void*, const void*, void*, int64_t, int,
                         hipStream_t);
void ce_fwd_launch(const void*, const int64_t*, float*, float*, int64_t, int,
                   int, int, int64_t, hipStream_t);
void ce_fwd_sharded_launch(const void*, const int64_t*, float*, float*, float*,
                           int64_t, int, int, int, int64_t, int64_t,
                           hipStream_t);
void ce_bwd_launch(const void*, const int64_t*, const float*, void*, float,
                   const float*, int64_t, int, int, int, int64_t, int64_t,
                   int, hipStream_t);
void adamw_launch(const void*, int, float, float, float, float, float, float,
                  float, hipStream_t);
void ln_fwd_launch(const void*, const void*, const void*, void*, float*,
                   float*, int64_t, int, float, hipStream_t);
void ln_bwd_launch(const void*, const void*, const void*, const float*,
                   const float*, void*, float*, float*, void*, void*, int,
                   int64_t, int, hipStream_t);
void gelu_fwd_launch(const void*, void*, int64_t, hipStream_t);
void gelu_bwd_launch(const void*, const void*, void*, int64_t, hipStream_t);
void embed_fwd_launch(const void*, const int64_t*, void*, int64_t, int,
                      int64_t, hipStream_t);
void embed_bwd_launch(const void*, const int64_t*, float*, void*, int64_t,
                      int, int64_t, hipStream_t);
void attn_fwd_launch(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, int, float, hipStream_t);
void attn_bwd_launch(const void*, const void*, const void*, const void*,
                     const void*, const float*, float*, void*, void*, void*,
                     int, int, int, int, int, float, hipStream_t);
}

namespace {

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_BF16_CONTIG(t)                                          \
  TORCH_CHECK((t).is_cuda(), #t " must be on the GPU");               \
  TORCH_CHECK((t).dtype() == torch::kBFloat16, #t " must be bf16");   \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// ---------------- rmsnorm ----------------
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(w);
  const int H = (int)x.size(-1);
  const int64_t nrows = x.numel() / H;
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({nrows}, x.options().dtype(torch::kFloat));
  rmsnorm_fwd_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(), rstd.data_ptr(),
                     nrows, H, (float)eps, cur_stream());
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(x);
  const int H = (int)x.size(-1);
  const int64_t nrows = x.numel() / H;
  const int nblocks = (int)std::min<int64_t>(nrows, 2048);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto dw_partial =
      torch::empty({(int64_t)nblocks, (int64_t)H}, x.options().dtype(torch::kFloat));
  rmsnorm_bwd_launch(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     rstd.data_ptr(), dx.data_ptr(),
                     dw_partial.data_ptr<float>(), dw.data_ptr(), nblocks,
                     nrows, H, cur_stream());
  return {dx, dw};
}

// ---------------- rope ----------------
torch::Tensor rope(torch::Tensor x, torch::Tensor cos, torch::Tensor sin,
                   c10::optional<torch::Tensor> positions, bool backward) {
  CHECK_BF16_CONTIG(x);
  TORCH_CHECK(x.dim() == 4, "rope expects [B,S,H,D]");
  const int S = (int)x.size(1), H = (int)x.size(2), D = (int)x.size(3);
  TORCH_CHECK(cos.is_contiguous() && sin.is_contiguous());
  TORCH_CHECK(cos.dtype() == torch::kFloat);
  const int* pos_ptr = nullptr;
  if (positions.has_value()) {
    TORCH_CHECK(positions->dtype() == torch::kInt && positions->is_contiguous());
    TORCH_CHECK(positions->numel() == S, "positions must be [S]");
    pos_ptr = positions->data_ptr<int>();
  } else {
    TORCH_CHECK(cos.size(0) >= S, "rope table shorter than sequence");
  }
  auto y = torch::empty_like(x);
  rope_launch(x.data_ptr(), y.data_ptr(), cos.data_ptr<float>(),
              sin.data_ptr<float>(), pos_ptr, x.numel() / D, S, H, D,
              backward ? 1 : 0, cur_stream());
  return y;
}

// ---------------- fused residual add + rmsnorm ----------------
std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor res,
                                           torch::Tensor delta,
                                           torch::Tensor w, double eps) {
  CHECK_BF16_CONTIG(res);
  CHECK_BF16_CONTIG(delta);
  CHECK_BF16_CONTIG(w);
  const int H = (int)res.size(-1);
  TORCH_CHECK(H % 8 == 0 && H <= 16384, "add_rmsnorm: unsupported H");
  const int64_t nrows = res.numel() / H;
  auto res_out = torch::empty_like(res);
  auto y = torch::empty_like(res);
  auto rstd = torch::empty({nrows}, res.options().dtype(torch::kFloat));
  add_rmsnorm_fwd_launch(res.data_ptr(), delta.data_ptr(), w.data_ptr(),
                         res_out.data_ptr(), y.data_ptr(), rstd.data_ptr(),
                         nrows, H, (float)eps, cur_stream());
  return {y, res_out, rstd};
}

std::vector<torch::Tensor> add_rmsnorm_bwd(torch::Tensor dy,
                                           torch::Tensor dres_out,
                                           torch::Tensor res_out,
                                           torch::Tensor w,
                                           torch::Tensor rstd) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(dres_out);
  CHECK_BF16_CONTIG(res_out);
  const int H = (int)res_out.size(-1);
  const int64_t nrows = res_out.numel() / H;
  const int nblocks = (int)std::min<int64_t>(nrows, 2048);
  auto dx = torch::empty_like(res_out);
  auto dw = torch::empty_like(w);
  auto dw_partial =
      torch::empty({(int64_t)nblocks, (int64_t)H},
                   res_out.options().dtype(torch::kFloat));
  add_rmsnorm_bwd_launch(dy.data_ptr(), dres_out.data_ptr(),
                         res_out.data_ptr(), w.data_ptr(), rstd.data_ptr(),
                         dx.data_ptr(), dw_partial.data_ptr<float>(),
                         dw.data_ptr(), nblocks, nrows, H, cur_stream());
  return {dx, dw};
}

// ---------------- fused qkv split + rope ----------------
std::vector<torch::Tensor> qkv_rope_fwd(torch::Tensor qkv, torch::Tensor cos,
                                        torch::Tensor sin,
                                        c10::optional<torch::Tensor> positions,
                                        int64_t Hq, int64_t Hkv, int64_t D) {
  CHECK_BF16_CONTIG(qkv);
  TORCH_CHECK(qkv.dim() == 3, "qkv_rope expects [B,S,W]");
  const int64_t B = qkv.size(0), S = qkv.size(1);
  TORCH_CHECK(qkv.size(2) == (Hq + 2 * Hkv) * D, "packed width mismatch");
  TORCH_CHECK(D % 16 == 0,
              "qkv_rope requires head_dim % 16 == 0 (8-pair vectorized "
              "kernel); use the unfused rope path otherwise");
  const int* pos_ptr = nullptr;
  if (positions.has_value()) {
    TORCH_CHECK(positions->dtype() == torch::kInt &&
                positions->is_contiguous() && positions->numel() == S);
    pos_ptr = positions->data_ptr<int>();
  } else {
    TORCH_CHECK(cos.size(0) >= S, "rope table shorter than sequence");
  }
  auto q = torch::empty({B, S, Hq, D}, qkv.options());
  auto k = torch::empty({B, S, Hkv, D}, qkv.options());
  auto v = torch::empty({B, S, Hkv, D}, qkv.options());
  qkv_rope_fwd_launch(qkv.data_ptr(), q.data_ptr(), k.data_ptr(),
                      v.data_ptr(), cos.data_ptr<float>(),
                      sin.data_ptr<float>(), pos_ptr, B * S, (int)S, (int)Hq,
                      (int)Hkv, (int)D, cur_stream());
  return {q, k, v};
}

torch::Tensor qkv_rope_bwd(torch::Tensor dq, torch::Tensor dk,
                           torch::Tensor dv, torch::Tensor cos,
                           torch::Tensor sin,
                           c10::optional<torch::Tensor> positions) {
  CHECK_BF16_CONTIG(dq);
  CHECK_BF16_CONTIG(dk);
  CHECK_BF16_CONTIG(dv);
  const int64_t B = dq.size(0), S = dq.size(1);
  const int64_t Hq = dq.size(2), Hkv = dk.size(2), D = dq.size(3);
  TORCH_CHECK(D % 16 == 0, "qkv_rope requires head_dim % 16 == 0");
  const int* pos_ptr = nullptr;
  if (positions.has_value()) pos_ptr = positions->data_ptr<int>();
  auto dqkv = torch::empty({B, S, (Hq + 2 * Hkv) * D}, dq.options());
  qkv_rope_bwd_launch(dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                      dqkv.data_ptr(), cos.data_ptr<float>(),
                      sin.data_ptr<float>(), pos_ptr, B * S, (int)S, (int)Hq,
                      (int)Hkv, (int)D, cur_stream());
  return dqkv;
}

// ---------------- silu_mul ----------------
torch::Tensor silu_mul_fwd(torch::Tensor gu) {
  CHECK_BF16_CONTIG(gu);
  const int I2 = (int)gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "packed gate_up dim must be a multiple of 16");
  const int I = I2 / 2;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto y = torch::empty(sizes, gu.options());
  silu_mul_fwd_launch(gu.data_ptr(), y.data_ptr(), gu.numel() / I2, I,
                      cur_stream());
  return y;
}

torch::Tensor silu_mul_bwd(torch::Tensor dy, torch::Tensor gu) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(gu);
  const int I2 = (int)gu.size(-1);
  const int I = I2 / 2;
  auto dgu = torch::empty_like(gu);
  silu_mul_bwd_launch(dy.data_ptr(), gu.data_ptr(), dgu.data_ptr(),
                      gu.numel() / I2, I, cur_stream());
  return dgu;
}

// ---------------- embedding ----------------
torch::Tensor embedding_fwd(torch::Tensor table, torch::Tensor ids) {
  CHECK_BF16_CONTIG(table);
  TORCH_CHECK(ids.dtype() == torch::kLong && ids.is_contiguous());
  const int64_t V = table.size(0);
  const int H = (int)table.size(1);
  auto sizes = ids.sizes().vec();
  sizes.push_back(H);
  auto out = torch::empty(sizes, table.options());
  embed_fwd_launch(table.data_ptr(), ids.data_ptr<int64_t>(),
                   out.data_ptr(), ids.numel(), H, V, cur_stream());
  return out;
}

torch::Tensor embedding_bwd(torch::Tensor dy, torch::Tensor ids, int64_t V) {
  CHECK_BF16_CONTIG(dy);
  TORCH_CHECK(ids.dtype() == torch::kLong && ids.is_contiguous());
  const int H = (int)dy.size(-1);
  TORCH_CHECK(dy.numel() == ids.numel() * H, "dy/ids shape mismatch");
  auto acc = torch::empty({V, (int64_t)H},
                          dy.options().dtype(torch::kFloat));
  auto dtable = torch::empty({V, (int64_t)H}, dy.options());
  embed_bwd_launch(dy.data_ptr(), ids.data_ptr<int64_t>(),
                   acc.data_ptr<float>(), dtable.data_ptr(), ids.numel(), H,
                   V, cur_stream());
  return dtable;
}

// ---------------- layernorm + gelu (GPT-2 ops) ----------------
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(w);
  CHECK_BF16_CONTIG(b);
  const int H = (int)x.size(-1);
  const int64_t nrows = x.numel() / H;
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat);
  auto mu = torch::empty({nrows}, opts);
  auto rstd = torch::empty({nrows}, opts);
  ln_fwd_launch(x.data_ptr(), w.data_ptr(), b.data_ptr(), y.data_ptr(),
                mu.data_ptr<float>(), rstd.data_ptr<float>(), nrows, H,
                (float)eps, cur_stream());
  return {y, mu, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mu,
                                         torch::Tensor rstd) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(x);
  const int H = (int)x.size(-1);
  const int64_t nrows = x.numel() / H;
  const int nblocks = (int)std::min<int64_t>(nrows, 2048);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto db = torch::empty_like(w);
  auto opts = x.options().dtype(torch::kFloat);
  auto dw_partial = torch::empty({(int64_t)nblocks, (int64_t)H}, opts);
  auto db_partial = torch::empty({(int64_t)nblocks, (int64_t)H}, opts);
  ln_bwd_launch(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                mu.data_ptr<float>(), rstd.data_ptr<float>(), dx.data_ptr(),
                dw_partial.data_ptr<float>(), db_partial.data_ptr<float>(),
                dw.data_ptr(), db.data_ptr(), nblocks, nrows, H,
                cur_stream());
  return {dx, dw, db};
}

torch::Tensor gelu_fwd(torch::Tensor x) {
  CHECK_BF16_CONTIG(x);
  TORCH_CHECK(x.numel() % 8 == 0, "gelu kernel needs numel % 8 == 0");
  auto y = torch::empty_like(x);
  gelu_fwd_launch(x.data_ptr(), y.data_ptr(), x.numel(), cur_stream());
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(x);
  auto dx = torch::empty_like(x);
  gelu_bwd_launch(dy.data_ptr(), x.data_ptr(), dx.data_ptr(), x.numel(),
                  cur_stream());
  return dx;
}

// ---------------- cross entropy ----------------
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  int64_t S_out, int64_t ignore_index) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(labels.dtype() == torch::kLong && labels.is_contiguous());
  TORCH_CHECK(logits.dim() == 3, "logits must be [B,S,V]");
  const int B = (int)logits.size(0), S = (int)logits.size(1),
            V = (int)logits.size(2);
  TORCH_CHECK(V % 8 == 0, "vocab must be padded to a multiple of 8");
  const int64_t nrows = (int64_t)B * S_out;
  auto loss = torch::empty({nrows}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({nrows}, logits.options().dtype(torch::kFloat));
  ce_fwd_launch(logits.data_ptr(), labels.data_ptr<int64_t>(),
                loss.data_ptr<float>(), lse.data_ptr<float>(), nrows, S,
                (int)S_out, V, ignore_index, cur_stream());
  return {loss, lse};
}

std::vector<torch::Tensor> ce_fwd_sharded(torch::Tensor logits,
                                          torch::Tensor labels, int64_t S_out,
                                          int64_t vocab_start,
                                          int64_t ignore_index) {
  CHECK_BF16_CONTIG(logits);
  const int B = (int)logits.size(0), S = (int)logits.size(1),
            V = (int)logits.size(2);
  const int64_t nrows = (int64_t)B * S_out;
  auto opts = logits.options().dtype(torch::kFloat);
  auto maxout = torch::empty({nrows}, opts);
  auto sumout = torch::empty({nrows}, opts);
  auto gathered = torch::empty({nrows}, opts);
  ce_fwd_sharded_launch(logits.data_ptr(), labels.data_ptr<int64_t>(),
                        maxout.data_ptr<float>(), sumout.data_ptr<float>(),
                        gathered.data_ptr<float>(), nrows, S, (int)S_out, V,
                        vocab_start, ignore_index, cur_stream());
  return {maxout, sumout, gathered};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor lse, double scale, int64_t S_out,
                     int64_t vocab_start, int64_t ignore_index, bool sharded,
                     c10::optional<torch::Tensor> scale_t = c10::nullopt) {
  CHECK_BF16_CONTIG(logits);
  const int B = (int)logits.size(0), S = (int)logits.size(1),
            V = (int)logits.size(2);
  const int64_t nrows = (int64_t)B * S_out;
  const float* sp = nullptr;
  if (scale_t.has_value()) {
    TORCH_CHECK(scale_t->dtype() == torch::kFloat && scale_t->is_cuda() &&
                scale_t->numel() == 1, "scale_t must be a cuda f32 scalar");
    sp = scale_t->data_ptr<float>();
  }
  // all rows with S_out == S are loss rows (the kernel writes every one);
  // otherwise the skipped rows must stay zero
  auto dlogits = (S_out == S) ? torch::empty_like(logits)
                              : torch::zeros_like(logits);
  ce_bwd_launch(logits.data_ptr(), labels.data_ptr<int64_t>(),
                lse.data_ptr<float>(), dlogits.data_ptr(), (float)scale, sp,
                nrows, S, (int)S_out, V, vocab_start, ignore_index,
                sharded ? 1 : 0, cur_stream());
  return dlogits;
}

// ---------------- adamw ----------------
void adamw_step(torch::Tensor descs, int64_t nchunks, double lr, double beta1,
                double beta2, double eps, double wd, int64_t step) {
  TORCH_CHECK(descs.is_cuda() && descs.dtype() == torch::kLong &&
              descs.is_contiguous());
  const float bc1 = 1.0f - std::pow((float)beta1, (float)step);
  const float bc2 = 1.0f - std::pow((float)beta2, (float)step);
  adamw_launch(descs.data_ptr(), (int)nchunks, (float)lr, (float)beta1,
               (float)beta2, (float)eps, (float)wd, 1.0f / bc1, 1.0f / bc2,
               cur_stream());
}

// ---------------- attention ----------------
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, double scale) {
  CHECK_BF16_CONTIG(q);
  CHECK_BF16_CONTIG(k);
  CHECK_BF16_CONTIG(v);
  TORCH_CHECK(q.dim() == 4, "q must be [B,S,Hq,D]");
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({(int64_t)B, (int64_t)Hq, (int64_t)S},
                          q.options().dtype(torch::kFloat));
  attn_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr<float>(), B, S, Hq, Hkv, D, (float)scale,
                  cur_stream());
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    double scale) {
  CHECK_BF16_CONTIG(dout);
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({(int64_t)B, (int64_t)Hq, (int64_t)S},
                            q.options().dtype(torch::kFloat));
  attn_bwd_launch(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                  o.data_ptr(), lse.data_ptr<float>(),
                  delta.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                  dv.data_ptr(), B, S, Hq, Hkv, D, (float)scale,
                  cur_stream());
  return {dq, dk, dv};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rope", &rope);
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("add_rmsnorm_bwd", &add_rmsnorm_bwd);
  m.def("qkv_rope_fwd", &qkv_rope_fwd);
  m.def("qkv_rope_bwd", &qkv_rope_bwd);
  m.def("silu_mul_fwd", &silu_mul_fwd);
  m.def("silu_mul_bwd", &silu_mul_bwd);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_fwd_sharded", &ce_fwd_sharded);
  m.def("ce_bwd", &ce_bwd, py::arg("logits"), py::arg("labels"),
        py::arg("lse"), py::arg("scale"), py::arg("S_out"),
        py::arg("vocab_start"), py::arg("ignore_index"),
        py::arg("sharded"), py::arg("scale_t") = c10::nullopt);
  m.def("adamw_step", &adamw_step);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
}
