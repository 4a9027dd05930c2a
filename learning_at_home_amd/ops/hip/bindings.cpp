// Python bindings for the MI355X kernel suite (torch extension).
// Thin layer: dtype/contiguity checks, output allocation, stream plumbing.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime.h>

extern "C" {
void lah_batched_gemm_nt(const void*, const void*, void*, const void*, void*, int, int,
                         int, int, int, hipStream_t);
void lah_batched_gemm_tn(const void*, const void*, void*, int, int, int, int, int,
                         hipStream_t);
void lah_batched_gemm_tn_sgd_sr(const void*, const void*, void*, void*, int, int, int,
                                int, float, float, unsigned, hipStream_t);
void lah_flash_attn_fwd(const void*, const void*, const void*, void*, void*, void*, int, int,
                        hipStream_t);
void lah_flash_attn_bwd(const void*, const void*, const void*, const void*, const void*,
                        const void*, void*, void*, void*, void*, void*, void*, int, int,
                        hipStream_t);
void lah_gating_topk_fwd(const void*, void*, void*, void*, int, int, int, hipStream_t);
void lah_gating_topk_bwd(const void*, const void*, const void*, const void*, const void*,
                         void*, int, int, int, hipStream_t);
void lah_assign_slots(const void*, void*, void*, void*, int, int, int, int, hipStream_t);
void lah_dispatch_gather(const void*, const void*, void*, int, int, int, int, hipStream_t);
void lah_dispatch_bwd(const void*, const void*, const void*, void*, int, int, int, int,
                      hipStream_t);
void lah_combine_fwd(const void*, const void*, const void*, const void*, void*, int, int,
                     int, int, hipStream_t);
void lah_combine_bwd(const void*, const void*, const void*, const void*, const void*,
                     const void*, void*, void*, int, int, int, int, hipStream_t);
void lah_zero_padding(const void*, void*, int, int, int, hipStream_t);
void lah_sgd_step(void*, const void*, void*, void*, long, float, float, float,
                  hipStream_t);
void lah_sgd_step_t(void*, const void*, void*, void*, int, int, int, float, float,
                    hipStream_t);
void lah_sgd_step_sr_t(void*, const void*, void*, int, int, int, float, float,
                       unsigned, hipStream_t);
void lah_transpose_bf16(const void*, void*, int, int, int, hipStream_t);
void lah_ce_rowstats(const void*, const void*, void*, void*, int, long, int, int,
                     hipStream_t);
void lah_ce_bwd(const void*, const void*, const void*, void*, float, int, long, int,
                int, hipStream_t);
void lah_colsum(const void*, void*, int, int, int, hipStream_t);
}

namespace {

hipStream_t cur_stream() { return at::hip::getCurrentHIPStream().stream(); }

#define CHECK_BF16(t) TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")
#define CHECK_F32(t) TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be fp32")
#define CHECK_I32(t) TORCH_CHECK((t).scalar_type() == at::kInt, #t " must be int32")
#define CHECK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")
#define CHECK_CUDA(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

// C[e] = A[e] @ B[e]^T with fused epilogue; see ffn_gemm.hip
torch::Tensor batched_gemm_nt(torch::Tensor A, torch::Tensor B,
                              c10::optional<torch::Tensor> bias,
                              c10::optional<torch::Tensor> pre, int64_t epi) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_CONTIG(A);
  CHECK_BF16(B); CHECK_CONTIG(B);
  int E = A.size(0), M = A.size(1), K = A.size(2);
  int N = B.size(1);
  TORCH_CHECK(B.size(0) == E && B.size(2) == K, "A/B shape mismatch");
  TORCH_CHECK(N % 128 == 0, "N must be a multiple of 128");
  TORCH_CHECK(K % 64 == 0, "K must be a multiple of 64");
  auto C = torch::empty({E, M, N}, A.options());
  const void* bias_p = nullptr;
  void* pre_p = nullptr;
  if (bias.has_value()) { CHECK_F32(*bias); CHECK_CONTIG(*bias); bias_p = bias->data_ptr(); }
  if (epi == 2) {  // BIAS_GELU stores pre-activation
    TORCH_CHECK(pre.has_value(), "epi=2 requires a pre-activation output tensor");
  }
  if (pre.has_value()) { CHECK_BF16(*pre); CHECK_CONTIG(*pre); pre_p = pre->data_ptr(); }
  lah_batched_gemm_nt(A.data_ptr(), B.data_ptr(), C.data_ptr(), bias_p, pre_p, E, M, N, K,
                      (int)epi, cur_stream());
  return C;
}

// C[e,i,j] (fp32, written in place) = sum_m A[e,m,i] * B[e,m,j]
void batched_gemm_tn(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                     bool accumulate) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_CONTIG(A);
  CHECK_BF16(B); CHECK_CONTIG(B); CHECK_F32(C); CHECK_CONTIG(C);
  int E = A.size(0), M = A.size(1), I = A.size(2), J = B.size(2);
  TORCH_CHECK(B.size(0) == E && B.size(1) == M, "A/B shape mismatch");
  TORCH_CHECK(C.size(0) == E && C.size(1) == I && C.size(2) == J, "C shape mismatch");
  TORCH_CHECK(M % 64 == 0, "M must be a multiple of 64");
  TORCH_CHECK(I % 128 == 0 && J % 128 == 0, "I/J must be multiples of 128");
  lah_batched_gemm_tn(A.data_ptr(), B.data_ptr(), C.data_ptr(), E, M, I, J,
                      accumulate ? 1 : 0, cur_stream());
}

// fused dW + stochastic-rounding SGD: W[e] -= lr * (A[e]^T @ B[e] + wd*W[e]),
// SR-rounded in place, with the transposed bf16 copy refreshed in the same
// pass — replaces batched_gemm_tn + sgd_step_sr_t for the single-backward-
// per-step regime (see ffn_gemm_tn.hip)
void fused_dw_sgd_sr(torch::Tensor A, torch::Tensor B, torch::Tensor W,
                     torch::Tensor WT, double lr, double weight_decay,
                     int64_t seed) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_CONTIG(A);
  CHECK_BF16(B); CHECK_CONTIG(B);
  CHECK_BF16(W); CHECK_CONTIG(W); CHECK_BF16(WT); CHECK_CONTIG(WT);
  int E = A.size(0), M = A.size(1), I = A.size(2), J = B.size(2);
  TORCH_CHECK(B.size(0) == E && B.size(1) == M, "A/B shape mismatch");
  TORCH_CHECK(W.size(0) == E && W.size(1) == I && W.size(2) == J, "W shape mismatch");
  TORCH_CHECK(WT.size(0) == E && WT.size(1) == J && WT.size(2) == I, "WT shape mismatch");
  TORCH_CHECK(M % 64 == 0, "M must be a multiple of 64");
  TORCH_CHECK(I % 256 == 0 && J % 256 == 0,
              "fused dW+SGD needs I/J multiples of 256 (use the unfused path)");
  lah_batched_gemm_tn_sgd_sr(A.data_ptr(), B.data_ptr(), W.data_ptr(), WT.data_ptr(),
                             E, M, I, J, (float)lr, (float)weight_decay,
                             (unsigned)seed, cur_stream());
}

// chunked lm_head CE row stats: (lse, picked) per row, one logits read
std::vector<torch::Tensor> ce_rowstats(torch::Tensor logits, torch::Tensor labels,
                                       int64_t ignore_index) {
  CHECK_CUDA(logits); CHECK_CONTIG(logits);
  bool bf16 = logits.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 || logits.scalar_type() == at::kFloat, "logits must be bf16 or fp32");
  CHECK_I32(labels); CHECK_CONTIG(labels);
  int T = logits.size(0);
  long V = logits.size(1);
  TORCH_CHECK(labels.size(0) == T, "labels/logits mismatch");
  TORCH_CHECK(V % 4 == 0, "V must be a multiple of 4");
  auto lse = torch::empty({T}, logits.options().dtype(at::kFloat));
  auto picked = torch::empty({T}, logits.options().dtype(at::kFloat));
  lah_ce_rowstats(logits.data_ptr(), labels.data_ptr(), lse.data_ptr(),
                  picked.data_ptr(), T, V, (int)ignore_index, (int)bf16,
                  cur_stream());
  return {lse, picked};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                     double gscale, int64_t ignore_index) {
  CHECK_CUDA(logits); CHECK_CONTIG(logits);
  bool bf16 = logits.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 || logits.scalar_type() == at::kFloat, "logits must be bf16 or fp32");
  CHECK_I32(labels); CHECK_CONTIG(labels); CHECK_F32(lse); CHECK_CONTIG(lse);
  int T = logits.size(0);
  long V = logits.size(1);
  TORCH_CHECK(V % 4 == 0, "V must be a multiple of 4");
  auto d = torch::empty({T, V}, logits.options().dtype(at::kBFloat16));
  lah_ce_bwd(logits.data_ptr(), labels.data_ptr(), lse.data_ptr(), d.data_ptr(),
             (float)gscale, T, V, (int)ignore_index, (int)bf16, cur_stream());
  return d;
}

std::vector<torch::Tensor> gating_topk_fwd(torch::Tensor logits, int64_t k) {
  CHECK_CUDA(logits); CHECK_F32(logits); CHECK_CONTIG(logits);
  int T = logits.size(0), E = logits.size(1);
  TORCH_CHECK(E <= 4096, "gating kernel supports up to 4096 experts, got ", E);
  TORCH_CHECK(k == 1 || k == 2 || k == 4 || k == 8,
              "gating kernel supports k in {1,2,4,8}, got ", k);
  auto probs = torch::empty({T, E}, logits.options().dtype(at::kBFloat16));
  auto topk_idx = torch::empty({T, k}, logits.options().dtype(at::kInt));
  auto topk_w = torch::empty({T, k}, logits.options());
  lah_gating_topk_fwd(logits.data_ptr(), probs.data_ptr(), topk_idx.data_ptr(),
                      topk_w.data_ptr(), T, E, (int)k, cur_stream());
  return {probs, topk_idx, topk_w};
}

torch::Tensor gating_topk_bwd(torch::Tensor probs, torch::Tensor topk_idx,
                              torch::Tensor topk_w, torch::Tensor d_topk_w,
                              c10::optional<torch::Tensor> aux_g) {
  CHECK_BF16(probs); CHECK_I32(topk_idx); CHECK_F32(topk_w); CHECK_F32(d_topk_w);
  int T = probs.size(0), E = probs.size(1), k = topk_idx.size(1);
  TORCH_CHECK(E <= 4096, "gating kernel supports up to 4096 experts, got ", E);
  auto d_logits = torch::empty({T, E}, topk_w.options());
  const void* aux_p = nullptr;
  if (aux_g.has_value()) { CHECK_F32(*aux_g); CHECK_CONTIG(*aux_g); aux_p = aux_g->data_ptr(); }
  lah_gating_topk_bwd(probs.data_ptr(), topk_idx.data_ptr(), topk_w.data_ptr(),
                      d_topk_w.contiguous().data_ptr(), aux_p, d_logits.data_ptr(), T, E, k,
                      cur_stream());
  return d_logits;
}

std::vector<torch::Tensor> assign_slots(torch::Tensor topk_idx, int64_t E, int64_t C) {
  CHECK_CUDA(topk_idx); CHECK_I32(topk_idx); CHECK_CONTIG(topk_idx);
  int T = topk_idx.size(0), k = topk_idx.size(1);
  auto pos = torch::empty({T, k}, topk_idx.options());
  auto counts = torch::zeros({E}, topk_idx.options());
  auto inv = torch::full({E, C}, -1, topk_idx.options());
  lah_assign_slots(topk_idx.data_ptr(), pos.data_ptr(), counts.data_ptr(), inv.data_ptr(),
                   T, k, (int)C, (int)E, cur_stream());
  return {pos, counts, inv};
}

torch::Tensor dispatch_gather(torch::Tensor x, torch::Tensor inv, int64_t k) {
  CHECK_BF16(x); CHECK_CONTIG(x); CHECK_I32(inv); CHECK_CONTIG(inv);
  int E = inv.size(0), C = inv.size(1), D = x.size(1);
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8");
  auto xbuf = torch::empty({E, C, D}, x.options());
  lah_dispatch_gather(x.data_ptr(), inv.data_ptr(), xbuf.data_ptr(), E, C, D, (int)k,
                      cur_stream());
  return xbuf;
}

torch::Tensor dispatch_bwd(torch::Tensor dxbuf, torch::Tensor topk_idx, torch::Tensor pos,
                           int64_t T) {
  CHECK_BF16(dxbuf); CHECK_CONTIG(dxbuf);
  int C = dxbuf.size(1), D = dxbuf.size(2), k = topk_idx.size(1);
  auto dx = torch::empty({T, D}, dxbuf.options());
  lah_dispatch_bwd(dxbuf.data_ptr(), topk_idx.data_ptr(), pos.data_ptr(), dx.data_ptr(),
                   (int)T, D, k, C, cur_stream());
  return dx;
}

torch::Tensor combine_fwd(torch::Tensor ybuf, torch::Tensor topk_idx, torch::Tensor pos,
                          torch::Tensor w) {
  CHECK_BF16(ybuf); CHECK_CONTIG(ybuf); CHECK_F32(w);
  int C = ybuf.size(1), D = ybuf.size(2);
  int T = topk_idx.size(0), k = topk_idx.size(1);
  auto y = torch::empty({T, D}, ybuf.options());
  lah_combine_fwd(ybuf.data_ptr(), topk_idx.data_ptr(), pos.data_ptr(), w.data_ptr(),
                  y.data_ptr(), T, D, k, C, cur_stream());
  return y;
}

std::vector<torch::Tensor> combine_bwd(torch::Tensor dy, torch::Tensor ybuf,
                                       torch::Tensor y, torch::Tensor topk_idx,
                                       torch::Tensor pos, torch::Tensor w,
                                       torch::Tensor inv) {
  CHECK_BF16(dy); CHECK_BF16(ybuf); CHECK_BF16(y);
  int E = ybuf.size(0), C = ybuf.size(1), D = ybuf.size(2);
  int T = topk_idx.size(0), k = topk_idx.size(1);
  auto dybuf = torch::empty({E, C, D}, ybuf.options());
  auto dw = torch::empty({T, k}, w.options());
  // padding rows (inv < 0) are zero-filled; surviving rows written by scatter
  lah_zero_padding(inv.data_ptr(), dybuf.data_ptr(), E, C, D, cur_stream());
  lah_combine_bwd(dy.contiguous().data_ptr(), ybuf.data_ptr(), y.data_ptr(),
                  topk_idx.data_ptr(), pos.data_ptr(), w.data_ptr(), dybuf.data_ptr(),
                  dw.data_ptr(), T, D, k, C, cur_stream());
  return {dybuf, dw};
}

void sgd_step(torch::Tensor master, torch::Tensor grad, c10::optional<torch::Tensor> momentum,
              torch::Tensor out_bf16, double lr, double mu, double weight_decay) {
  CHECK_F32(master); CHECK_F32(grad); CHECK_BF16(out_bf16);
  CHECK_CONTIG(master); CHECK_CONTIG(grad); CHECK_CONTIG(out_bf16);
  long n = master.numel();
  TORCH_CHECK(grad.numel() == n && out_bf16.numel() == n, "size mismatch");
  TORCH_CHECK(n % 4 == 0, "numel must be a multiple of 4");
  void* mom_p = nullptr;
  if (momentum.has_value()) { CHECK_F32(*momentum); mom_p = momentum->data_ptr(); }
  lah_sgd_step(master.data_ptr(), grad.data_ptr(), mom_p, out_bf16.data_ptr(), n, (float)lr,
               (float)mu, (float)weight_decay, cur_stream());
}

void sgd_step_t(torch::Tensor master, torch::Tensor grad, torch::Tensor out,
                torch::Tensor outT, double lr, double weight_decay) {
  CHECK_F32(master); CHECK_F32(grad); CHECK_BF16(out); CHECK_BF16(outT);
  CHECK_CONTIG(master); CHECK_CONTIG(grad); CHECK_CONTIG(out); CHECK_CONTIG(outT);
  int E = master.size(0), R = master.size(1), C = master.size(2);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0, "dims must be multiples of 64");
  TORCH_CHECK(outT.size(1) == C && outT.size(2) == R, "outT must be [E, C, R]");
  lah_sgd_step_t(master.data_ptr(), grad.data_ptr(), out.data_ptr(), outT.data_ptr(),
                 E, R, C, (float)lr, (float)weight_decay, cur_stream());
}

void sgd_step_sr_t(torch::Tensor param, torch::Tensor grad, torch::Tensor paramT,
                   double lr, double weight_decay, int64_t seed) {
  CHECK_BF16(param); CHECK_F32(grad); CHECK_BF16(paramT);
  CHECK_CONTIG(param); CHECK_CONTIG(grad); CHECK_CONTIG(paramT);
  int E = param.size(0), R = param.size(1), C = param.size(2);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0, "dims must be multiples of 64");
  lah_sgd_step_sr_t(param.data_ptr(), grad.data_ptr(), paramT.data_ptr(), E, R, C,
                    (float)lr, (float)weight_decay, (unsigned)seed, cur_stream());
}

torch::Tensor transpose_bf16(torch::Tensor in) {
  CHECK_BF16(in); CHECK_CONTIG(in); CHECK_CUDA(in);
  int E = in.size(0), R = in.size(1), C = in.size(2);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0, "transpose dims must be multiples of 64");
  auto out = torch::empty({E, C, R}, in.options());
  lah_transpose_bf16(in.data_ptr(), out.data_ptr(), E, R, C, cur_stream());
  return out;
}

torch::Tensor colsum(torch::Tensor d) {
  CHECK_BF16(d); CHECK_CONTIG(d);
  int E = d.size(0), M = d.size(1), C = d.size(2);
  auto db = torch::empty({E, C}, d.options().dtype(at::kFloat));
  lah_colsum(d.data_ptr(), db.data_ptr(), E, M, C, cur_stream());
  return db;
}

}  // namespace


// causal flash-attention forward (experimental): q,k,v [B,H,S,64] bf16
static void check_attn_args(const torch::Tensor& q, const torch::Tensor& k,
                            const torch::Tensor& v) {
  CHECK_CUDA(q); CHECK_BF16(q); CHECK_CONTIG(q);
  CHECK_BF16(k); CHECK_CONTIG(k); CHECK_BF16(v); CHECK_CONTIG(v);
  TORCH_CHECK(q.dim() == 4 && q.size(3) == 64, "q must be [B, H, S, 64]");
  TORCH_CHECK(q.sizes() == k.sizes() && q.sizes() == v.sizes(), "q/k/v shape mismatch");
  TORCH_CHECK(q.size(2) % 128 == 0, "seq_len must be a multiple of 128");
}

torch::Tensor flash_attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v) {
  check_attn_args(q, k, v);
  auto o = torch::empty_like(q);
  auto vt = torch::empty_like(v);  // [BH, 64, S] scratch via torch allocator
  lah_flash_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), vt.data_ptr(),
                     o.data_ptr(), nullptr, q.size(0) * q.size(1), q.size(2),
                     cur_stream());
  return o;
}

// (o, lse) — the training entry point; lse feeds the recompute-based backward
std::vector<torch::Tensor> flash_attn_fwd_lse(torch::Tensor q, torch::Tensor k,
                                              torch::Tensor v) {
  check_attn_args(q, k, v);
  auto o = torch::empty_like(q);
  auto vt = torch::empty_like(v);
  auto lse = torch::empty({q.size(0), q.size(1), q.size(2)},
                          q.options().dtype(at::kFloat));
  lah_flash_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), vt.data_ptr(),
                     o.data_ptr(), lse.data_ptr(), q.size(0) * q.size(1), q.size(2),
                     cur_stream());
  return {o, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor dout,
                                          torch::Tensor lse, torch::Tensor d) {
  check_attn_args(q, k, v);
  CHECK_BF16(dout); CHECK_CONTIG(dout);
  CHECK_F32(lse); CHECK_CONTIG(lse); CHECK_F32(d); CHECK_CONTIG(d);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto qt = torch::empty_like(q);   // transposed-operand scratch
  auto kt = torch::empty_like(k);
  auto dot = torch::empty_like(dout);
  lah_flash_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), dout.data_ptr(),
                     lse.data_ptr(), d.data_ptr(), qt.data_ptr(), kt.data_ptr(),
                     dot.data_ptr(), dq.data_ptr(), dk.data_ptr(),
                     dv.data_ptr(), q.size(0) * q.size(1), q.size(2), cur_stream());
  return {dq, dk, dv};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("batched_gemm_nt", &batched_gemm_nt, "batched NT GEMM with fused epilogue",
        py::arg("A"), py::arg("B"), py::arg("bias") = py::none(),
        py::arg("pre") = py::none(), py::arg("epi") = 0);
  m.def("batched_gemm_tn", &batched_gemm_tn, py::arg("A"), py::arg("B"), py::arg("C"),
        py::arg("accumulate") = false);
  m.def("fused_dw_sgd_sr", &fused_dw_sgd_sr, py::arg("A"), py::arg("B"), py::arg("W"),
        py::arg("WT"), py::arg("lr"), py::arg("weight_decay") = 0.0,
        py::arg("seed") = 0);
  m.def("ce_rowstats", &ce_rowstats, py::arg("logits"), py::arg("labels"),
        py::arg("ignore_index") = -100);
  m.def("ce_bwd", &ce_bwd, py::arg("logits"), py::arg("labels"), py::arg("lse"),
        py::arg("gscale"), py::arg("ignore_index") = -100);
  m.def("flash_attn_fwd", &flash_attn_fwd);
  m.def("flash_attn_fwd_lse", &flash_attn_fwd_lse);
  m.def("flash_attn_bwd", &flash_attn_bwd);
  m.def("gating_topk_fwd", &gating_topk_fwd);
  m.def("gating_topk_bwd", &gating_topk_bwd, py::arg("probs"), py::arg("topk_idx"),
        py::arg("topk_w"), py::arg("d_topk_w"), py::arg("aux_g") = py::none());
  m.def("assign_slots", &assign_slots);
  m.def("dispatch_gather", &dispatch_gather);
  m.def("dispatch_bwd", &dispatch_bwd);
  m.def("combine_fwd", &combine_fwd);
  m.def("combine_bwd", &combine_bwd);
  m.def("sgd_step", &sgd_step, py::arg("master"), py::arg("grad"),
        py::arg("momentum"), py::arg("out_bf16"),
        py::arg("lr") = 0.01, py::arg("mu") = 0.0, py::arg("weight_decay") = 0.0);
  m.def("sgd_step_t", &sgd_step_t, py::arg("master"), py::arg("grad"), py::arg("out"),
        py::arg("outT"), py::arg("lr") = 0.01, py::arg("weight_decay") = 0.0);
  m.def("sgd_step_sr_t", &sgd_step_sr_t, py::arg("param"), py::arg("grad"),
        py::arg("paramT"), py::arg("lr") = 0.01, py::arg("weight_decay") = 0.0,
        py::arg("seed") = 0);
  m.def("transpose_bf16", &transpose_bf16);
  m.def("colsum", &colsum);
}
