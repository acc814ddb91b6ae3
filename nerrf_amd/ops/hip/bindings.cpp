// Torch extension bindings for the nerrf-amd CDNA4 kernels.
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "mcts_params.h"

namespace nerrf {
void launch_mcts(const float*, const float*, const float*, float, float,
                 const PlannerParamsDev&, int, int, int*, float*, int*, float*,
                 hipStream_t);
void launch_eval_plans(const float*, const float*, const float*, float, float,
                       const PlannerParamsDev&, const int*, int, int, float*,
                       hipStream_t);
void launch_gather_mean_fwd(const void*, const long*, const float*, void*, int,
                            int, int, bool, hipStream_t);
void launch_gather_mean_bwd(const void*, const long*, const float*, float*,
                            int, int, int, bool, hipStream_t);
void launch_gather_mean_bwd_csr(const void*, const long*, const long*,
                                const float*, float*, long, int, bool,
                                hipStream_t);
void launch_lstm_pointwise_fwd(const void*, const void*, const void*,
                               const void*, const void*, const float*, void*,
                               void*, void*, long, int, long, long, long,
                               bool, hipStream_t);
void launch_lstm_pointwise_bwd(const void*, const void*, const void*,
                               const void*, const void*, const float*, void*,
                               void*, void*, float*, long, int, long, long,
                               bool, hipStream_t);
void launch_lstm_step_fused(const void*, const void*, const void*, const void*,
                            const void*, const float*, void*, void*, void*,
                            int, bool, hipStream_t);
void launch_lstm_rec_fwd(const void*, const void*, const void*, const void*,
                         const void*, const float*, void*, void*, void*, int,
                         long, long, long, hipStream_t);
void launch_lstm_rec_bwd(const void*, const void*, const void*, const void*,
                         const void*, const void*, const float*, void*, void*,
                         void*, int, long, long, hipStream_t);
void launch_rec_gemm_fwd(const void*, const void*, void*, long, long, long,
                         hipStream_t);
void launch_rec_gemm_dgrad(const void*, const void*, const void*, void*,
                           long, long, hipStream_t);
void launch_proj_fwd_dual(const void*, const void*, const void*, void*, void*,
                          long, long, hipStream_t);
void launch_proj_dgrad_dual(const void*, const void*, const void*, const void*,
                            void*, long, hipStream_t);
void launch_proj_wgrad(const void*, const void*, const void*, float*, float*,
                       long, long, int, hipStream_t);
void launch_event_scatter(const long*, const long*, const signed char*,
                          const float*, const int*, float*, int*, int*, long,
                          hipStream_t);
void launch_feature_assemble(const float*, const int*, const int*,
                             const float*, const float*, const float*,
                             const unsigned char*, const signed char*, float*,
                             float, int, hipStream_t);
void launch_sage_layer_fwd(const void*, const long*, const float*, const void*,
                           const void*, const void*, const void*, const void*,
                           void*, int, int, hipStream_t);
void launch_sage_ln_act_fwd(const void*, const void*, const void*, const void*,
                            const void*, void*, void*, float*, unsigned char*,
                            long, float, unsigned, hipStream_t);
void launch_sage_ln_act_bwd(const void*, const void*, const float*,
                            const void*, void*, float*, float*, long, float,
                            unsigned, hipStream_t);
}  // namespace nerrf

namespace {

bool is_bf16(const torch::Tensor& t) {
  TORCH_CHECK(
      t.scalar_type() == torch::kBFloat16 || t.scalar_type() == torch::kFloat32,
      "nerrf kernels support bf16/fp32, got ", t.scalar_type());
  return t.scalar_type() == torch::kBFloat16;
}

void check_gpu_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// 2D tensor whose rows may be strided (last dim contiguous) — the dual-
// direction LSTM reads gate slabs / writes hidden halves inside wider
// concatenated buffers.
long row_stride_checked(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dim() == 2 && t.stride(1) == 1, name,
              " must be 2D with contiguous rows");
  return t.stride(0);
}

torch::Tensor gather_mean_fwd(torch::Tensor h, torch::Tensor idx,
                              torch::Tensor w) {
  check_gpu_contig(h, "h");
  check_gpu_contig(idx, "idx");
  check_gpu_contig(w, "w");
  TORCH_CHECK(idx.scalar_type() == torch::kInt64, "idx must be int64");
  const int n = idx.size(0);
  const int k = idx.size(1);
  const int dim = h.size(1);
  TORCH_CHECK(k <= 64, "fanout K must be <= 64 (wave-resident), got ", k);
  TORCH_CHECK(dim <= 512, "feature dim must be <= 512, got ", dim);
  auto wf = w.scalar_type() == torch::kFloat32 ? w : w.to(torch::kFloat32);
  auto out = torch::empty({n, dim}, h.options());
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_gather_mean_fwd(h.data_ptr(), idx.data_ptr<long>(),
                                wf.data_ptr<float>(), out.data_ptr(), n, dim,
                                k, is_bf16(h), stream.stream());
  return out;
}

torch::Tensor gather_mean_bwd(torch::Tensor grad_out, torch::Tensor idx,
                              torch::Tensor w, long num_nodes) {
  check_gpu_contig(grad_out, "grad_out");
  check_gpu_contig(idx, "idx");
  const int n = idx.size(0);
  const int k = idx.size(1);
  const int dim = grad_out.size(1);
  auto wf = w.scalar_type() == torch::kFloat32 ? w.contiguous()
                                               : w.to(torch::kFloat32).contiguous();
  auto ws = torch::zeros({num_nodes, dim},
                         grad_out.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_gather_mean_bwd(grad_out.data_ptr(), idx.data_ptr<long>(),
                                wf.data_ptr<float>(), ws.data_ptr<float>(), n,
                                dim, k, is_bf16(grad_out), stream.stream());
  return ws.to(grad_out.scalar_type());
}

torch::Tensor gather_mean_bwd_csr(torch::Tensor grad_out,
                                  torch::Tensor rev_dst,
                                  torch::Tensor rev_src, torch::Tensor rev_w,
                                  long num_nodes) {
  check_gpu_contig(grad_out, "grad_out");
  check_gpu_contig(rev_dst, "rev_dst");
  check_gpu_contig(rev_src, "rev_src");
  check_gpu_contig(rev_w, "rev_w");
  const int dim = grad_out.size(1);
  const long n_entries = rev_dst.numel();
  auto ws = torch::zeros({num_nodes, (long)dim},
                         grad_out.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_gather_mean_bwd_csr(
      grad_out.data_ptr(), rev_dst.data_ptr<long>(), rev_src.data_ptr<long>(),
      rev_w.data_ptr<float>(), ws.data_ptr<float>(), n_entries, dim,
      is_bf16(grad_out), stream.stream());
  return ws.to(grad_out.scalar_type());
}

// Writes into caller-provided (h_out, c_out, gates_act) so the sequence loop
// can target time-major buffers directly (no per-step allocation or copy).
void lstm_pointwise_fwd(torch::Tensor hg, torch::Tensor xg, torch::Tensor bias,
                        torch::Tensor c_prev, torch::Tensor h_prev,
                        torch::Tensor mask, torch::Tensor h_out,
                        torch::Tensor c_out, torch::Tensor gates_act) {
  check_gpu_contig(hg, "hg");
  const long xg_stride = row_stride_checked(xg, "xg");
  check_gpu_contig(c_prev, "c_prev");
  const long hprev_stride = row_stride_checked(h_prev, "h_prev");
  const long hout_stride = row_stride_checked(h_out, "h_out");
  check_gpu_contig(c_out, "c_out");
  const bool want_gates = gates_act.numel() > 0;
  if (want_gates) check_gpu_contig(gates_act, "gates_act");
  const long batch = c_prev.size(0);
  const int hdim = c_prev.size(1);
  TORCH_CHECK(hg.size(1) == 4 * hdim && xg.size(1) == 4 * hdim,
              "hg/xg must be [B, 4H]");
  TORCH_CHECK(bias.numel() == 4 * hdim, "bias must be [4H]");
  const float* mask_ptr = nullptr;
  torch::Tensor mf;
  if (mask.numel() > 0) {
    mf = mask.scalar_type() == torch::kFloat32 ? mask.contiguous()
                                               : mask.to(torch::kFloat32).contiguous();
    TORCH_CHECK(mf.numel() == batch, "mask must be [B]");
    mask_ptr = mf.data_ptr<float>();
  }
  auto bc = bias.contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_lstm_pointwise_fwd(
      hg.data_ptr(), xg.data_ptr(), bc.data_ptr(), c_prev.data_ptr(),
      h_prev.data_ptr(), mask_ptr, h_out.data_ptr(), c_out.data_ptr(),
      want_gates ? gates_act.data_ptr() : nullptr, batch, hdim, xg_stride,
      hout_stride, hprev_stride, is_bf16(hg), stream.stream());
}

// grad_out_t (may be empty): this timestep's dL/dh, folded in-kernel so the
// sequence backward needs no separate elementwise add per step.
// bias_accum (optional, may be empty): f32 [4H] accumulator the kernel
// folds sum_b(grad_gates) into (vectorised path only) — callers then skip
// the separate whole-tensor gg.sum(0) reduction.  Returns true when the
// fused accumulation ran.
bool lstm_pointwise_bwd(torch::Tensor grad_h, torch::Tensor grad_out_t,
                        torch::Tensor grad_c, torch::Tensor gates_act,
                        torch::Tensor c_prev, torch::Tensor mask,
                        torch::Tensor grad_gates, torch::Tensor grad_c_prev,
                        torch::Tensor grad_h_pass, torch::Tensor bias_accum) {
  check_gpu_contig(grad_h, "grad_h");
  check_gpu_contig(grad_c, "grad_c");
  check_gpu_contig(gates_act, "gates_act");
  check_gpu_contig(c_prev, "c_prev");
  const long gg_stride = row_stride_checked(grad_gates, "grad_gates");
  check_gpu_contig(grad_c_prev, "grad_c_prev");
  check_gpu_contig(grad_h_pass, "grad_h_pass");
  const long batch = c_prev.size(0);
  const int hdim = c_prev.size(1);
  const float* mask_ptr = nullptr;
  torch::Tensor mf;
  if (mask.numel() > 0) {
    mf = mask.scalar_type() == torch::kFloat32 ? mask.contiguous()
                                               : mask.to(torch::kFloat32).contiguous();
    mask_ptr = mf.data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  const void* got = grad_out_t.numel() ? grad_out_t.data_ptr() : nullptr;
  long gout_stride = hdim;
  if (grad_out_t.numel()) gout_stride = row_stride_checked(grad_out_t, "grad_out_t");
  float* bias_ptr = nullptr;
  const int v = is_bf16(grad_h) ? 8 : 4;
  const bool vec = hdim % v == 0 && gout_stride % v == 0 && gg_stride % v == 0;
  if (bias_accum.numel() > 0 && vec) {
    check_gpu_contig(bias_accum, "bias_accum");
    TORCH_CHECK(bias_accum.scalar_type() == torch::kFloat32 &&
                    bias_accum.numel() == 64 * 4 * hdim,
                "bias_accum must be f32 [64, 4H] (striped replicas)");
    bias_ptr = bias_accum.data_ptr<float>();
  }
  nerrf::launch_lstm_pointwise_bwd(
      grad_h.data_ptr(), got, grad_c.data_ptr(), gates_act.data_ptr(),
      c_prev.data_ptr(), mask_ptr, grad_gates.data_ptr(),
      grad_c_prev.data_ptr(), grad_h_pass.data_ptr(), bias_ptr, batch, hdim,
      gout_stride, gg_stride, is_bf16(grad_h), stream.stream());
  return bias_ptr != nullptr;
}

// Fused recurrent step (bf16, H == 256): h_prev @ W_hh^T + LSTM pointwise
// in one launch — the [B, 4H] pre-activation slab never touches HBM.
void lstm_rec_fwd(torch::Tensor h_prev, torch::Tensor w_hh, torch::Tensor xg,
                  torch::Tensor bias, torch::Tensor c_prev, torch::Tensor mask,
                  torch::Tensor h_out, torch::Tensor c_out,
                  torch::Tensor gates_act) {
  const long hprev_stride = row_stride_checked(h_prev, "h_prev");
  const long xg_stride = row_stride_checked(xg, "xg");
  const long hout_stride = row_stride_checked(h_out, "h_out");
  check_gpu_contig(w_hh, "w_hh");
  check_gpu_contig(c_prev, "c_prev");
  check_gpu_contig(c_out, "c_out");
  const bool want_gates = gates_act.numel() > 0;
  if (want_gates) check_gpu_contig(gates_act, "gates_act");
  for (auto* t : {&h_prev, &w_hh, &xg, &c_prev, &h_out, &c_out}) {
    TORCH_CHECK(t->scalar_type() == torch::kBFloat16,
                "lstm_rec_fwd is bf16-only");
  }
  const long batch = c_prev.size(0);
  TORCH_CHECK(c_prev.size(1) == 256 && h_prev.size(1) == 256 &&
                  xg.size(1) == 1024 && w_hh.size(0) == 1024 &&
                  w_hh.size(1) == 256,
              "lstm_rec_fwd requires H=256");
  const float* mask_ptr = nullptr;
  torch::Tensor mf;
  if (mask.numel() > 0) {
    mf = mask.scalar_type() == torch::kFloat32 ? mask.contiguous()
                                               : mask.to(torch::kFloat32).contiguous();
    mask_ptr = mf.data_ptr<float>();
  }
  auto bc = bias.contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_lstm_rec_fwd(
      h_prev.data_ptr(), w_hh.data_ptr(), xg.data_ptr(), bc.data_ptr(),
      c_prev.data_ptr(), mask_ptr, h_out.data_ptr(), c_out.data_ptr(),
      want_gates ? gates_act.data_ptr() : nullptr, (int)batch, hprev_stride,
      xg_stride, hout_stride, stream.stream());
}

// Fused recurrent backward (bf16, H == 256): gate grads + grad_c_prev +
// grad_h = ghp + grad_gates @ W_hh in one launch.
void lstm_rec_bwd(torch::Tensor grad_h, torch::Tensor grad_out_t,
                  torch::Tensor grad_c, torch::Tensor gates_act,
                  torch::Tensor c_prev, torch::Tensor w_hh_t,
                  torch::Tensor mask, torch::Tensor grad_gates,
                  torch::Tensor grad_c_prev, torch::Tensor grad_h_out) {
  check_gpu_contig(grad_h, "grad_h");
  check_gpu_contig(grad_c, "grad_c");
  check_gpu_contig(gates_act, "gates_act");
  check_gpu_contig(c_prev, "c_prev");
  check_gpu_contig(w_hh_t, "w_hh_t");
  const long gg_stride = row_stride_checked(grad_gates, "grad_gates");
  check_gpu_contig(grad_c_prev, "grad_c_prev");
  check_gpu_contig(grad_h_out, "grad_h_out");
  for (auto* t : {&grad_h, &grad_c, &gates_act, &c_prev, &w_hh_t}) {
    TORCH_CHECK(t->scalar_type() == torch::kBFloat16,
                "lstm_rec_bwd is bf16-only");
  }
  const long batch = c_prev.size(0);
  TORCH_CHECK(c_prev.size(1) == 256 && gates_act.size(1) == 1024 &&
                  w_hh_t.size(0) == 256 && w_hh_t.size(1) == 1024,
              "lstm_rec_bwd requires H=256 and w_hh_t = W_hh^T contiguous");
  const float* mask_ptr = nullptr;
  torch::Tensor mf;
  if (mask.numel() > 0) {
    mf = mask.scalar_type() == torch::kFloat32 ? mask.contiguous()
                                               : mask.to(torch::kFloat32).contiguous();
    mask_ptr = mf.data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  const void* got = grad_out_t.numel() ? grad_out_t.data_ptr() : nullptr;
  long gout_stride = 256;
  if (grad_out_t.numel()) gout_stride = row_stride_checked(grad_out_t, "grad_out_t");
  nerrf::launch_lstm_rec_bwd(
      grad_h.data_ptr(), got, grad_c.data_ptr(), gates_act.data_ptr(),
      c_prev.data_ptr(), w_hh_t.data_ptr(), mask_ptr, grad_gates.data_ptr(),
      grad_c_prev.data_ptr(), grad_h_out.data_ptr(), (int)batch, gout_stride,
      gg_stride, stream.stream());
}

// Fully-fused MFMA step (bf16, H == 256). Writes into caller buffers.
void lstm_step_fused(torch::Tensor h_prev, torch::Tensor w_hh, torch::Tensor xg,
                     torch::Tensor bias, torch::Tensor c_prev,
                     torch::Tensor mask, torch::Tensor h_out,
                     torch::Tensor c_out, torch::Tensor gates_act, bool raw) {
  for (auto* t : {&h_prev, &w_hh, &xg, &c_prev, &h_out, &c_out, &gates_act}) {
    check_gpu_contig(*t, "lstm_step_fused arg");
    TORCH_CHECK(t->scalar_type() == torch::kBFloat16,
                "lstm_step_fused is bf16-only");
  }
  const int batch = h_prev.size(0);
  TORCH_CHECK(h_prev.size(1) == 256 && xg.size(1) == 1024,
              "lstm_step_fused requires H=256 (got h ", h_prev.size(1), ")");
  TORCH_CHECK(w_hh.dim() == 3 && w_hh.size(0) == 8 && w_hh.size(1) == 1024 &&
                  w_hh.size(2) == 32,
              "w must be k-tiled [8,1024,32] (w_hh.reshape(1024,8,32)"
              ".permute(1,0,2).contiguous())");
  const float* mask_ptr = nullptr;
  torch::Tensor mf;
  if (mask.numel() > 0) {
    mf = mask.scalar_type() == torch::kFloat32 ? mask.contiguous()
                                               : mask.to(torch::kFloat32).contiguous();
    mask_ptr = mf.data_ptr<float>();
  }
  auto bc = bias.contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_lstm_step_fused(
      h_prev.data_ptr(), w_hh.data_ptr(), xg.data_ptr(), bc.data_ptr(),
      c_prev.data_ptr(), mask_ptr, h_out.data_ptr(), c_out.data_ptr(),
      gates_act.data_ptr(), batch, raw, stream.stream());
}

// Recurrent-step GEMM: c = a @ w^T with [M,256] x [1024,256] (bf16; A and C
// may be row-strided column slabs of wider buffers).
void rec_gemm_fwd(torch::Tensor a, torch::Tensor w, torch::Tensor c) {
  const long a_stride = row_stride_checked(a, "a");
  const long c_stride = row_stride_checked(c, "c");
  check_gpu_contig(w, "w");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16, "rec_gemm_fwd is bf16-only");
  TORCH_CHECK(a.size(1) == 256 && w.size(0) == 1024 && w.size(1) == 256 &&
                  c.size(1) == 1024 && c.size(0) == a.size(0),
              "rec_gemm_fwd requires [M,256] x [1024,256] -> [M,1024]");
  TORCH_CHECK(a_stride % 8 == 0 && c_stride % 8 == 0,
              "rec_gemm_fwd needs 16-B aligned rows");
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_rec_gemm_fwd(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                             a.size(0), a_stride, c_stride, stream.stream());
}

// Recurrence backward dgrad: c = a @ wt^T + d with [M,1024] x [256,1024]
// (wt = W_hh^T contiguous; d optional addend, pass an empty tensor to skip).
void rec_gemm_dgrad(torch::Tensor a, torch::Tensor wt, torch::Tensor d,
                    torch::Tensor c) {
  const long a_stride = row_stride_checked(a, "a");
  check_gpu_contig(wt, "wt");
  check_gpu_contig(c, "c");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16, "rec_gemm_dgrad is bf16-only");
  TORCH_CHECK(a.size(1) == 1024 && wt.size(0) == 256 && wt.size(1) == 1024 &&
                  c.size(1) == 256 && c.size(0) == a.size(0),
              "rec_gemm_dgrad requires [M,1024] x [256,1024] -> [M,256]");
  TORCH_CHECK(a_stride % 8 == 0, "rec_gemm_dgrad needs 16-B aligned rows");
  const bool has_d = d.numel() > 0;
  if (has_d) {
    check_gpu_contig(d, "d");
    TORCH_CHECK(d.sizes() == c.sizes(), "addend shape must match output");
  }
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_rec_gemm_dgrad(a.data_ptr(), wt.data_ptr(),
                               has_d ? d.data_ptr() : nullptr, c.data_ptr(),
                               a.size(0), a_stride, stream.stream());
}

// Dual-direction LSTM input projection: c1 = a @ w1^T, c2 = a @ w2^T in one
// launch (A staged in LDS once; K=512, N=1024 per direction, bf16).
void proj_fwd_dual(torch::Tensor a, torch::Tensor w1, torch::Tensor w2,
                   torch::Tensor c1, torch::Tensor c2) {
  const long a_stride = row_stride_checked(a, "a");
  check_gpu_contig(w1, "w1");
  check_gpu_contig(c1, "c1");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16, "proj_fwd_dual is bf16-only");
  TORCH_CHECK(a.size(1) == 512 && w1.size(0) == 1024 && w1.size(1) == 512 &&
                  c1.size(1) == 1024,
              "proj_fwd_dual requires [M,512] x [1024,512]");
  const bool dual = w2.numel() > 0;
  if (dual) {
    check_gpu_contig(w2, "w2");
    check_gpu_contig(c2, "c2");
    TORCH_CHECK(w2.sizes() == w1.sizes() && c2.sizes() == c1.sizes(),
                "dual shapes must match");
  }
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_proj_fwd_dual(
      a.data_ptr(), w1.data_ptr(), dual ? w2.data_ptr() : nullptr,
      c1.data_ptr(), dual ? c2.data_ptr() : nullptr, a.size(0), a_stride,
      stream.stream());
}

// Dual-direction input-projection dgrad: c = a1 @ w1t^T + a2 @ w2t^T
// (w*t = W^T contiguous [512, 1024]).
void proj_dgrad_dual(torch::Tensor a1, torch::Tensor a2, torch::Tensor w1t,
                     torch::Tensor w2t, torch::Tensor c) {
  check_gpu_contig(a1, "a1");
  check_gpu_contig(w1t, "w1t");
  check_gpu_contig(c, "c");
  TORCH_CHECK(a1.scalar_type() == torch::kBFloat16, "proj_dgrad_dual is bf16-only");
  TORCH_CHECK(a1.size(1) == 1024 && w1t.size(0) == 512 && w1t.size(1) == 1024 &&
                  c.size(1) == 512,
              "proj_dgrad_dual requires [M,1024] x [512,1024]");
  const bool dual = a2.numel() > 0;
  if (dual) {
    check_gpu_contig(a2, "a2");
    check_gpu_contig(w2t, "w2t");
  }
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_proj_dgrad_dual(
      a1.data_ptr(), dual ? a2.data_ptr() : nullptr, w1t.data_ptr(),
      dual ? w2t.data_ptr() : nullptr, c.data_ptr(), a1.size(0),
      stream.stream());
}

// Dual-direction input-projection weight grads: dW_d = g_d^T @ x for both
// directions in one launch (transposed LDS tiles, f32 atomic partials).
std::vector<torch::Tensor> proj_wgrad(torch::Tensor g1, torch::Tensor g2,
                                      torch::Tensor x, long n_mchunks) {
  // g1/g2 may be column slabs of one [M, 2048] cat-layout grad (rows
  // contiguous, shared row stride)
  const long g_stride = row_stride_checked(g1, "g1");
  check_gpu_contig(x, "x");
  TORCH_CHECK(g1.scalar_type() == torch::kBFloat16, "proj_wgrad is bf16-only");
  TORCH_CHECK(g1.size(1) == 1024 && x.size(1) == 512,
              "proj_wgrad requires g [M,1024], x [M,512]");
  const bool dual = g2.numel() > 0;
  if (dual)
    TORCH_CHECK(row_stride_checked(g2, "g2") == g_stride,
                "g1/g2 must share a row stride");
  auto dw1 = torch::zeros({1024, 512}, x.options().dtype(torch::kFloat32));
  auto dw2 = dual ? torch::zeros({1024, 512}, x.options().dtype(torch::kFloat32))
                  : torch::empty({0}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_proj_wgrad(g1.data_ptr(), dual ? g2.data_ptr() : nullptr,
                           x.data_ptr(), dw1.data_ptr<float>(),
                           dual ? dw2.data_ptr<float>() : nullptr, g1.size(0),
                           g_stride, (int)n_mchunks, stream.stream());
  return {dw1, dw2};
}

// GPU delta compaction: event columns -> per-node accumulators -> x [M, 32].
torch::Tensor event_features(torch::Tensor ev_file, torch::Tensor ev_proc,
                             torch::Tensor syscall_id, torch::Tensor nbytes,
                             torch::Tensor ts_ms, long m_nodes,
                             torch::Tensor in_deg, torch::Tensor out_deg,
                             torch::Tensor peer, torch::Tensor flags,
                             torch::Tensor node_kind, double span_s) {
  for (auto* t : {&ev_file, &ev_proc, &syscall_id, &nbytes, &ts_ms, &in_deg,
                  &out_deg, &peer, &flags, &node_kind})
    check_gpu_contig(*t, "event_features arg");
  const long n_events = ev_file.numel();
  auto opts_f = nbytes.options().dtype(torch::kFloat32);
  auto opts_i = nbytes.options().dtype(torch::kInt32);
  auto acc = torch::zeros({m_nodes, 13}, opts_f);
  auto t_first = torch::full({m_nodes}, (long)INT_MAX, opts_i);
  auto t_last = torch::full({m_nodes}, (long)INT_MIN, opts_i);
  auto x = torch::empty({m_nodes, 32}, opts_f);
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_event_scatter(
      ev_file.data_ptr<long>(), ev_proc.data_ptr<long>(),
      syscall_id.data_ptr<signed char>(), nbytes.data_ptr<float>(),
      ts_ms.data_ptr<int>(), acc.data_ptr<float>(), t_first.data_ptr<int>(),
      t_last.data_ptr<int>(), n_events, stream.stream());
  nerrf::launch_feature_assemble(
      acc.data_ptr<float>(), t_first.data_ptr<int>(), t_last.data_ptr<int>(),
      in_deg.data_ptr<float>(), out_deg.data_ptr<float>(),
      peer.data_ptr<float>(), flags.data_ptr<unsigned char>(),
      node_kind.data_ptr<signed char>(), x.data_ptr<float>(), (float)span_s,
      (int)m_nodes, stream.stream());
  return x;
}

// Fused GraphSAGE-T layer forward (inference): gather + dual MFMA GEMM +
// GELU + LayerNorm + residual in one launch.  bf16, D=128 only.
torch::Tensor sage_layer_fwd(torch::Tensor h, torch::Tensor nbr_idx,
                             torch::Tensor nbr_w, torch::Tensor w_self,
                             torch::Tensor w_nbr, torch::Tensor bias,
                             torch::Tensor gamma, torch::Tensor beta) {
  for (auto* t : {&h, &w_self, &w_nbr}) {
    check_gpu_contig(*t, "sage arg");
    TORCH_CHECK(t->scalar_type() == torch::kBFloat16, "sage_layer_fwd is bf16-only");
  }
  check_gpu_contig(nbr_idx, "nbr_idx");
  TORCH_CHECK(h.size(1) == 128 && w_self.size(0) == 128 && w_self.size(1) == 128 &&
                  w_nbr.size(0) == 128 && w_nbr.size(1) == 128,
              "sage_layer_fwd requires D=128");
  const int n = h.size(0);
  const int k = nbr_idx.size(1);
  TORCH_CHECK(k <= 64, "fanout must be <= 64");
  auto wf = nbr_w.scalar_type() == torch::kFloat32 ? nbr_w.contiguous()
                                                   : nbr_w.to(torch::kFloat32).contiguous();
  auto bc = bias.contiguous().to(torch::kBFloat16);
  auto gc = gamma.contiguous().to(torch::kBFloat16);
  auto be = beta.contiguous().to(torch::kBFloat16);
  auto out = torch::empty_like(h);
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_sage_layer_fwd(h.data_ptr(), nbr_idx.data_ptr<long>(),
                               wf.data_ptr<float>(), w_self.data_ptr(),
                               w_nbr.data_ptr(), bc.data_ptr(), gc.data_ptr(),
                               be.data_ptr(), out.data_ptr(), n, k,
                               stream.stream());
  return out;
}

// Fused GraphSAGE training-layer tail: y = h + LN(dropout(GELU(zs+zn)))*g+b
std::vector<torch::Tensor> sage_ln_act_fwd(torch::Tensor h, torch::Tensor zs,
                                           torch::Tensor zn, torch::Tensor gamma,
                                           torch::Tensor beta, double drop_p,
                                           long seed, bool want_mask) {
  for (auto* t : {&h, &zs, &zn}) {
    check_gpu_contig(*t, "sage_ln_act arg");
    TORCH_CHECK(t->scalar_type() == torch::kBFloat16, "sage_ln_act is bf16-only");
    TORCH_CHECK(t->size(1) == 128, "sage_ln_act requires D=128");
  }
  const long n = h.size(0);
  auto gc = gamma.contiguous().to(torch::kBFloat16);
  auto bc = beta.contiguous().to(torch::kBFloat16);
  auto y = torch::empty_like(h);
  auto s_save = torch::empty_like(h);
  auto stats = torch::empty({n, 2}, h.options().dtype(torch::kFloat32));
  torch::Tensor mask = want_mask
      ? torch::zeros({n, 128}, h.options().dtype(torch::kUInt8))
      : torch::empty({0}, h.options().dtype(torch::kUInt8));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_sage_ln_act_fwd(
      h.data_ptr(), zs.data_ptr(), zn.data_ptr(), gc.data_ptr(), bc.data_ptr(),
      y.data_ptr(), s_save.data_ptr(), stats.data_ptr<float>(),
      want_mask ? mask.data_ptr<unsigned char>() : nullptr, n, (float)drop_p,
      (unsigned)seed, stream.stream());
  return {y, s_save, stats, mask};
}

std::vector<torch::Tensor> sage_ln_act_bwd(torch::Tensor dy, torch::Tensor s_save,
                                           torch::Tensor stats, torch::Tensor gamma,
                                           double drop_p, long seed) {
  check_gpu_contig(dy, "dy");
  check_gpu_contig(s_save, "s_save");
  check_gpu_contig(stats, "stats");
  const long n = dy.size(0);
  auto gc = gamma.contiguous().to(torch::kBFloat16);
  auto dz = torch::empty_like(dy);
  const long grid = (n + 7) / 8;
  auto dg_part = torch::empty({grid, 128}, dy.options().dtype(torch::kFloat32));
  auto db_part = torch::empty({grid, 128}, dy.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_sage_ln_act_bwd(
      dy.data_ptr(), s_save.data_ptr(), stats.data_ptr<float>(), gc.data_ptr(),
      dz.data_ptr(), dg_part.data_ptr<float>(), db_part.data_ptr<float>(), n,
      (float)drop_p, (unsigned)seed, stream.stream());
  return {dz, dg_part.sum(0), db_part.sum(0)};
}

nerrf::PlannerParamsDev params_from_dict(const pybind11::dict& d) {
  nerrf::PlannerParamsDev p;
  p.n_groups = d["n_groups"].cast<int>();
  p.n_actions = p.n_groups + 3;  // STOP, KILL, RESTORE, revert g
  p.max_depth = d["max_depth"].cast<int>();
  p.sims_per_tree = d["sims_per_tree"].cast<int>();
  p.downtime_weight = d["downtime_weight"].cast<float>();
  p.revert_time_s = d["revert_time_s"].cast<float>();
  p.kill_time_s = d["kill_time_s"].cast<float>();
  p.fp_weight = d["fp_weight"].cast<float>();
  p.attack_rate_mbps = d["attack_rate_mbps"].cast<float>();
  p.horizon_s = d["horizon_s"].cast<float>();
  p.restore_time_s = d["restore_time_s"].cast<float>();
  p.restore_loss_mb = d["restore_loss_mb"].cast<float>();
  p.ucb_c = d["ucb_c"].cast<float>();
  p.seed = d["seed"].cast<unsigned>();
  TORCH_CHECK(p.n_groups <= 16, "n_groups must be <= 16");
  TORCH_CHECK(p.max_depth <= 16, "max_depth must be <= 16");
  return p;
}

std::vector<torch::Tensor> mcts_search(torch::Tensor gscore, torch::Tensor gmb,
                                       torch::Tensor gfiles, double proc_score,
                                       double remaining_clean_mb,
                                       pybind11::dict param_dict, long n_trees) {
  auto p = params_from_dict(param_dict);
  check_gpu_contig(gscore, "gscore");
  check_gpu_contig(gmb, "gmb");
  check_gpu_contig(gfiles, "gfiles");
  const int cap = p.sims_per_tree * p.max_depth + 2;
  auto opts_i = gscore.options().dtype(torch::kInt32);
  auto opts_f = gscore.options().dtype(torch::kFloat32);
  auto arena_i = torch::empty({n_trees, (long)(3 * cap + cap * p.n_actions)}, opts_i);
  auto arena_f = torch::empty({n_trees, (long)cap}, opts_f);
  auto root_visits = torch::zeros({n_trees, (long)p.n_actions}, opts_i);
  auto root_value = torch::zeros({n_trees, (long)p.n_actions}, opts_f);
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_mcts(gscore.data_ptr<float>(), gmb.data_ptr<float>(),
                     gfiles.data_ptr<float>(), (float)proc_score,
                     (float)remaining_clean_mb, p, (int)n_trees, cap,
                     arena_i.data_ptr<int>(), arena_f.data_ptr<float>(),
                     root_visits.data_ptr<int>(), root_value.data_ptr<float>(),
                     stream.stream());
  return {root_visits, root_value};
}

torch::Tensor mcts_eval_plans(torch::Tensor gscore, torch::Tensor gmb,
                              torch::Tensor gfiles, double proc_score,
                              double remaining_clean_mb,
                              pybind11::dict param_dict, torch::Tensor plans) {
  auto p = params_from_dict(param_dict);
  check_gpu_contig(plans, "plans");
  TORCH_CHECK(plans.scalar_type() == torch::kInt32, "plans must be int32");
  const int n_plans = plans.size(0);
  const int plan_len = plans.size(1);
  auto out = torch::empty({(long)n_plans}, gscore.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  nerrf::launch_eval_plans(gscore.data_ptr<float>(), gmb.data_ptr<float>(),
                           gfiles.data_ptr<float>(), (float)proc_score,
                           (float)remaining_clean_mb, p, plans.data_ptr<int>(),
                           n_plans, plan_len, out.data_ptr<float>(),
                           stream.stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sage_ln_act_fwd", &sage_ln_act_fwd,
        "fused GNN layer tail fwd (add+GELU+dropout+LN+residual)");
  m.def("sage_ln_act_bwd", &sage_ln_act_bwd, "fused GNN layer tail bwd");
  m.def("sage_layer_fwd", &sage_layer_fwd,
        "fused GraphSAGE-T layer forward (MFMA, inference)");
  m.def("event_features", &event_features,
        "GPU delta compaction: events -> per-node feature matrix");
  m.def("mcts_search", &mcts_search, "batched root-parallel MCTS");
  m.def("mcts_eval_plans", &mcts_eval_plans, "batch plan reward evaluation");
  m.def("gather_mean_fwd", &gather_mean_fwd, "weighted neighbor gather-mean");
  m.def("gather_mean_bwd", &gather_mean_bwd, "gather-mean backward");
  m.def("gather_mean_bwd_csr", &gather_mean_bwd_csr,
        "deterministic gather-mean backward over reverse CSR");
  m.def("lstm_pointwise_fwd", &lstm_pointwise_fwd, "fused LSTM gate pointwise fwd");
  m.def("lstm_rec_fwd", &lstm_rec_fwd,
        "fused recurrent GEMM + LSTM pointwise fwd (bf16, H=256)");
  m.def("lstm_rec_bwd", &lstm_rec_bwd,
        "fused LSTM gate grads + grad_h GEMM bwd (bf16, H=256)");
  m.def("rec_gemm_fwd", &rec_gemm_fwd,
        "recurrent-step GEMM [M,256]x[1024,256]^T (bf16, strided rows)");
  m.def("rec_gemm_dgrad", &rec_gemm_dgrad,
        "recurrence backward dgrad [M,1024]x[256,1024]^T + addend (bf16)");
  m.def("proj_fwd_dual", &proj_fwd_dual,
        "dual-direction LSTM input projection (A read once)");
  m.def("proj_dgrad_dual", &proj_dgrad_dual,
        "dual-direction input-projection dgrad (summed)");
  m.def("proj_wgrad", &proj_wgrad,
        "dual-direction input-projection weight grads (f32 partials)");
  m.def("lstm_step_fused", &lstm_step_fused, "fully-fused MFMA LSTM step (bf16, H=256)");
  m.def("lstm_pointwise_bwd", &lstm_pointwise_bwd, "fused LSTM gate pointwise bwd");
}
