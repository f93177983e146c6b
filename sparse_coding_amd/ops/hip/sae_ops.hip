// Torch bindings + launchers for the gfx950 SAE kernels (sae_kernels.hip).
// Built in-tree as sparse_coding_amd/ops/_sae_hip.so by ops/build.py.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "sae_kernels.hip"

#define CHECK_IN(t)                                           \
  TORCH_CHECK(t.is_cuda(), #t " must be on GPU");             \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous");   \
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, #t " must be fp32");

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static inline int cdiv(long a, long b) { return (int)((a + b - 1) / b); }

void row_norms(torch::Tensor W, torch::Tensor norms, torch::Tensor inv_norms, double eps) {
  CHECK_IN(W); CHECK_IN(norms); CHECK_IN(inv_norms);
  long rows = W.numel() / W.size(-1);
  int d = W.size(-1);
  dim3 grid(cdiv(rows, NTHREADS / WAVE));
  hipLaunchKernelGGL(k_row_norms, grid, dim3(NTHREADS), 0, cur_stream(),
                     W.data_ptr<float>(), norms.data_ptr<float>(),
                     inv_norms.data_ptr<float>(), (int)rows, d, (float)eps);
}

void enc_fwd(torch::Tensor x, torch::Tensor Wenc, torch::Tensor bias,
             c10::optional<torch::Tensor> inv_norms, torch::Tensor c_out,
             torch::Tensor loss_parts, torch::Tensor fired, int64_t mode,
             int64_t bk, bool prio,
             int64_t bn,
             c10::optional<torch::Tensor> act_scale,
             c10::optional<torch::Tensor> act_gain,
             c10::optional<torch::Tensor> u_out,
             c10::optional<torch::Tensor> dict_sizes,
             c10::optional<torch::Tensor> y_in,
             c10::optional<torch::Tensor> x_prev,
             c10::optional<torch::Tensor> x_out,
             c10::optional<torch::Tensor> mom) {
  CHECK_IN(x); CHECK_IN(Wenc); CHECK_IN(bias); CHECK_IN(c_out);
  CHECK_IN(loss_parts); CHECK_IN(fired);
  int M = Wenc.size(0), n = Wenc.size(1), d = Wenc.size(2);
  long x_mstride = 0;
  if (x.dim() == 3) {
    TORCH_CHECK(x.size(0) == M, "per-model x must be [M,B,d]");
    x_mstride = (long)x.size(1) * x.size(2);
  }
  int B = x.size(x.dim() - 2);
  TORCH_CHECK(d % 4 == 0 && n % 4 == 0, "d and n must be multiples of 4 (float4 staging)");
  const float* inv = nullptr;
  if (inv_norms.has_value()) {
    CHECK_IN(inv_norms.value());
    inv = inv_norms->data_ptr<float>();
  }
  const float* a_p = nullptr;
  const float* gn_p = nullptr;
  float* u_p = nullptr;
  if (mode == 2) {
    TORCH_CHECK(act_scale && act_gain && u_out, "mode 2 needs act_scale/act_gain/u_out");
    CHECK_IN(act_scale.value()); CHECK_IN(act_gain.value()); CHECK_IN(u_out.value());
    a_p = act_scale->data_ptr<float>();
    gn_p = act_gain->data_ptr<float>();
    u_p = u_out->data_ptr<float>();
  }
  const int* ds_p = nullptr;
  if (dict_sizes.has_value()) {
    TORCH_CHECK(dict_sizes->is_cuda() && dict_sizes->is_contiguous() &&
                dict_sizes->scalar_type() == torch::kInt32, "dict_sizes must be int32 GPU");
    ds_p = dict_sizes->data_ptr<int>();
  }
  const float* yin_p = nullptr;
  const float* xp_p = nullptr;
  float* xo_p = nullptr;
  const float* mom_p = nullptr;
  if (mode == 4 || mode == 5) {
    TORCH_CHECK(y_in, "modes 4/5 need y_in");
    CHECK_IN(y_in.value());
    yin_p = y_in->data_ptr<float>();
    if (mode == 4) {
      TORCH_CHECK(x_prev && x_out && mom && act_scale, "mode 4 needs x_prev/x_out/mom/act_scale(theta)/u_out");
      CHECK_IN(x_prev.value()); CHECK_IN(x_out.value()); CHECK_IN(mom.value());
      CHECK_IN(act_scale.value()); CHECK_IN(u_out.value());
      xp_p = x_prev->data_ptr<float>();
      xo_p = x_out->data_ptr<float>();
      mom_p = mom->data_ptr<float>();
      a_p = act_scale->data_ptr<float>();
      u_p = u_out->data_ptr<float>();
    }
  }
  dim3 grid(cdiv(n, bn == 256 ? 256 : BN), cdiv(B, BM), M);
  if (bn == 256)
    hipLaunchKernelGGL((k_enc_fwd_t<16, 4, 256>), grid, dim3(NTHREADS), 0, cur_stream(),
                       x.data_ptr<float>(), Wenc.data_ptr<float>(),
                       bias.data_ptr<float>(), inv, c_out.data_ptr<float>(),
                       loss_parts.data_ptr<float>(), fired.data_ptr<float>(),
                       B, d, n, (int)mode, prio ? 1 : 0, a_p, gn_p, u_p, ds_p, x_mstride,
                       yin_p, xp_p, xo_p, mom_p);
  else if (bk == 16)
    hipLaunchKernelGGL((k_enc_fwd_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       x.data_ptr<float>(), Wenc.data_ptr<float>(),
                       bias.data_ptr<float>(), inv, c_out.data_ptr<float>(),
                       loss_parts.data_ptr<float>(), fired.data_ptr<float>(),
                       B, d, n, (int)mode, prio ? 1 : 0, a_p, gn_p, u_p, ds_p, x_mstride,
                       yin_p, xp_p, xo_p, mom_p);
  else
    hipLaunchKernelGGL((k_enc_fwd_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       x.data_ptr<float>(), Wenc.data_ptr<float>(),
                       bias.data_ptr<float>(), inv, c_out.data_ptr<float>(),
                       loss_parts.data_ptr<float>(), fired.data_ptr<float>(),
                       B, d, n, (int)mode, prio ? 1 : 0, a_p, gn_p, u_p, ds_p, x_mstride,
                       yin_p, xp_p, xo_p, mom_p);
}

void dec_fwd(torch::Tensor c, torch::Tensor Wdec, torch::Tensor inv_norms,
             torch::Tensor x, torch::Tensor r_out, torch::Tensor loss_parts,
             int64_t bk, bool prio, int64_t bn) {
  CHECK_IN(c); CHECK_IN(Wdec); CHECK_IN(inv_norms); CHECK_IN(x);
  CHECK_IN(r_out); CHECK_IN(loss_parts);
  int M = Wdec.size(0), n = Wdec.size(1), d = Wdec.size(2);
  long x_mstride = 0;
  if (x.dim() == 3) {
    TORCH_CHECK(x.size(0) == M, "per-model x must be [M,B,d]");
    x_mstride = (long)x.size(1) * x.size(2);
  }
  int B = x.size(x.dim() - 2);
  dim3 grid(cdiv(d, bn == 256 ? 256 : BN), cdiv(B, BM), M);
  if (bn == 256)
    hipLaunchKernelGGL((k_dec_fwd_t<16, 4, 256>), grid, dim3(NTHREADS), 0, cur_stream(),
                       c.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), x.data_ptr<float>(),
                       r_out.data_ptr<float>(), loss_parts.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0, x_mstride);
  else if (bk == 16)
    hipLaunchKernelGGL((k_dec_fwd_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       c.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), x.data_ptr<float>(),
                       r_out.data_ptr<float>(), loss_parts.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0, x_mstride);
  else
    hipLaunchKernelGGL((k_dec_fwd_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       c.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), x.data_ptr<float>(),
                       r_out.data_ptr<float>(), loss_parts.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0, x_mstride);
}

void gc(torch::Tensor r, torch::Tensor Wdec, torch::Tensor inv_norms,
        torch::Tensor c, torch::Tensor l1_alpha, torch::Tensor gpre,
        torch::Tensor g_bias, int64_t bk, bool prio, int64_t gc_mode, int64_t bn) {
  CHECK_IN(r); CHECK_IN(Wdec); CHECK_IN(inv_norms); CHECK_IN(c);
  CHECK_IN(l1_alpha); CHECK_IN(gpre); CHECK_IN(g_bias);
  int M = Wdec.size(0), n = Wdec.size(1), d = Wdec.size(2);
  int B = r.size(1);
  dim3 grid(cdiv(n, bn == 256 ? 256 : BN), cdiv(B, BM), M);
  if (bn == 256)
    hipLaunchKernelGGL((k_gc_t<16, 4, 256>), grid, dim3(NTHREADS), 0, cur_stream(),
                       r.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), c.data_ptr<float>(),
                       l1_alpha.data_ptr<float>(), gpre.data_ptr<float>(),
                       g_bias.data_ptr<float>(), B, d, n, prio ? 1 : 0, (int)gc_mode);
  else if (bk == 16)
    hipLaunchKernelGGL((k_gc_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       r.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), c.data_ptr<float>(),
                       l1_alpha.data_ptr<float>(), gpre.data_ptr<float>(),
                       g_bias.data_ptr<float>(), B, d, n, prio ? 1 : 0, (int)gc_mode);
  else
    hipLaunchKernelGGL((k_gc_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       r.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), c.data_ptr<float>(),
                       l1_alpha.data_ptr<float>(), gpre.data_ptr<float>(),
                       g_bias.data_ptr<float>(), B, d, n, prio ? 1 : 0, (int)gc_mode);
}

void gc_thresh(torch::Tensor r, torch::Tensor Wdec, torch::Tensor inv_norms,
               torch::Tensor c, torch::Tensor u, torch::Tensor act_scale,
               torch::Tensor l1_alpha, torch::Tensor gpre,
               torch::Tensor g_gain, torch::Tensor g_scale,
               int64_t bk, bool prio) {
  CHECK_IN(r); CHECK_IN(Wdec); CHECK_IN(inv_norms); CHECK_IN(c); CHECK_IN(u);
  CHECK_IN(act_scale); CHECK_IN(l1_alpha); CHECK_IN(gpre);
  CHECK_IN(g_gain); CHECK_IN(g_scale);
  int M = Wdec.size(0), n = Wdec.size(1), d = Wdec.size(2);
  int B = r.size(1);
  dim3 grid(cdiv(n, BN), cdiv(B, BM), M);
  if (bk == 16)
    hipLaunchKernelGGL((k_gc_thresh_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       r.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), c.data_ptr<float>(),
                       u.data_ptr<float>(), act_scale.data_ptr<float>(),
                       l1_alpha.data_ptr<float>(), gpre.data_ptr<float>(),
                       g_gain.data_ptr<float>(), g_scale.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0);
  else
    hipLaunchKernelGGL((k_gc_thresh_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       r.data_ptr<float>(), Wdec.data_ptr<float>(),
                       inv_norms.data_ptr<float>(), c.data_ptr<float>(),
                       u.data_ptr<float>(), act_scale.data_ptr<float>(),
                       l1_alpha.data_ptr<float>(), gpre.data_ptr<float>(),
                       g_gain.data_ptr<float>(), g_scale.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0);
}

// gw[m] = beta * gw[m] + alpha * P[m]^T @ Q[m]; Q may be rank-shared [B, d]
void grad_w(torch::Tensor P, torch::Tensor Q, torch::Tensor gw,
            double alpha, double beta, int64_t bk, bool prio, int64_t bn) {
  CHECK_IN(P); CHECK_IN(Q); CHECK_IN(gw);
  int M = gw.size(0), n = gw.size(1), d = gw.size(2);
  int B;
  long p_stride, q_stride;
  TORCH_CHECK(P.dim() == 3, "P must be [M, B, n]");
  B = P.size(1);
  p_stride = (long)B * n;
  if (Q.dim() == 3) {
    q_stride = (long)B * d;
  } else {
    q_stride = 0;  // shared across models
  }
  dim3 grid(cdiv(d, bn == 256 ? 256 : BN), cdiv(n, BM), M);
  if (bn == 256)
    hipLaunchKernelGGL((k_grad_w_t<16, 4, 256>), grid, dim3(NTHREADS), 0, cur_stream(),
                       P.data_ptr<float>(), p_stride, Q.data_ptr<float>(),
                       q_stride, gw.data_ptr<float>(), (float)alpha,
                       (float)beta, B, n, d, prio ? 1 : 0);
  else if (bk == 16)
    hipLaunchKernelGGL((k_grad_w_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       P.data_ptr<float>(), p_stride, Q.data_ptr<float>(),
                       q_stride, gw.data_ptr<float>(), (float)alpha,
                       (float)beta, B, n, d, prio ? 1 : 0);
  else
    hipLaunchKernelGGL((k_grad_w_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       P.data_ptr<float>(), p_stride, Q.data_ptr<float>(),
                       q_stride, gw.data_ptr<float>(), (float)alpha,
                       (float)beta, B, n, d, prio ? 1 : 0);
}

void project_adam(torch::Tensor W, torch::Tensor gw, torch::Tensor norms,
                  torch::Tensor mu, torch::Tensor nu, torch::Tensor step_no,
                  long n_per_model, double lr, double b1, double b2,
                  double eps_adam, double eps_norm, bool project,
                  c10::optional<torch::Tensor> w_used, bool clamp_mask,
                  c10::optional<torch::Tensor> lr_mult) {
  CHECK_IN(W); CHECK_IN(gw); CHECK_IN(norms); CHECK_IN(mu); CHECK_IN(nu);
  CHECK_IN(step_no);
  const float* wu = nullptr;
  if (w_used.has_value()) {
    CHECK_IN(w_used.value());
    wu = w_used->data_ptr<float>();
  }
  long rows = W.numel() / W.size(-1);
  const float* lrm = nullptr;
  if (lr_mult.has_value()) {
    CHECK_IN(lr_mult.value());
    TORCH_CHECK(lr_mult->numel() == rows, "lr_mult must have one entry per row");
    lrm = lr_mult->data_ptr<float>();
  }
  int d = W.size(-1);
  dim3 grid(cdiv(rows, NTHREADS / WAVE));
  hipLaunchKernelGGL(k_project_adam, grid, dim3(NTHREADS), 0, cur_stream(),
                     W.data_ptr<float>(), gw.data_ptr<float>(),
                     norms.data_ptr<float>(), mu.data_ptr<float>(),
                     nu.data_ptr<float>(), step_no.data_ptr<float>(),
                     (int)rows, (int)n_per_model, d, (float)lr, (float)b1,
                     (float)b2, (float)eps_adam, (float)eps_norm,
                     project ? 1 : 0, wu, clamp_mask ? 1 : 0, lrm);
}

void bias_adam(torch::Tensor bias, torch::Tensor g_bias, torch::Tensor decay,
               torch::Tensor mu, torch::Tensor nu, torch::Tensor step_no,
               double lr, double b1, double b2, double eps_adam,
               c10::optional<torch::Tensor> lr_mult) {
  CHECK_IN(bias); CHECK_IN(g_bias); CHECK_IN(decay); CHECK_IN(mu);
  CHECK_IN(nu); CHECK_IN(step_no);
  int M = bias.size(0), n = bias.size(1);
  const float* lrm = nullptr;
  if (lr_mult.has_value()) {
    CHECK_IN(lr_mult.value());
    TORCH_CHECK(lr_mult->numel() == (long)M * n, "lr_mult must match bias shape");
    lrm = lr_mult->data_ptr<float>();
  }
  hipLaunchKernelGGL(k_bias_adam, dim3(M), dim3(NTHREADS), 0, cur_stream(),
                     bias.data_ptr<float>(), g_bias.data_ptr<float>(),
                     decay.data_ptr<float>(), mu.data_ptr<float>(),
                     nu.data_ptr<float>(), step_no.data_ptr<float>(), n,
                     (float)lr, (float)b1, (float)b2, (float)eps_adam, lrm);
}

void colsum(torch::Tensor X, torch::Tensor out, double alpha, bool absval) {
  CHECK_IN(X); CHECK_IN(out);
  TORCH_CHECK(X.dim() == 3, "colsum wants [M, B, n]");
  int M = X.size(0), B = X.size(1), n = X.size(2);
  TORCH_CHECK(out.numel() == (long)M * n, "out must be [M, n]");
  int nsplit = std::min(std::max(B / 256, 1), 16);
  hipMemsetAsync(out.data_ptr<float>(), 0, (size_t)M * n * sizeof(float),
                 cur_stream());
  dim3 grid(cdiv(n, 256), M, nsplit);
  hipLaunchKernelGGL(k_colsum, grid, dim3(256), 0, cur_stream(),
                     X.data_ptr<float>(), out.data_ptr<float>(), B, n,
                     (float)alpha, absval ? 1 : 0);
}

void transpose_scale(torch::Tensor src, torch::Tensor dst,
                     c10::optional<torch::Tensor> scale) {
  CHECK_IN(src); CHECK_IN(dst);
  int R, C, M;
  long src_ms = 0, dst_ms = 0, scale_ms = 0;
  if (src.dim() == 3) {
    M = src.size(0); R = src.size(1); C = src.size(2);
    src_ms = (long)R * C; dst_ms = (long)R * C;
  } else {
    M = 1; R = src.size(0); C = src.size(1);
  }
  const float* sc = nullptr;
  if (scale.has_value()) {
    CHECK_IN(scale.value());
    sc = scale->data_ptr<float>();
    scale_ms = (src.dim() == 3) ? R : 0;
  }
  dim3 grid(cdiv(C, 64), cdiv(R, 64), M);
  hipLaunchKernelGGL(k_transpose_scale, grid, dim3(256), 0, cur_stream(),
                     src.data_ptr<float>(), dst.data_ptr<float>(), sc,
                     R, C, src_ms, dst_ms, scale_ms);
}

void enc_fwd2(torch::Tensor xT, torch::Tensor WT, torch::Tensor bias,
              torch::Tensor c_out, torch::Tensor loss_parts,
              torch::Tensor fired, int64_t mode, int64_t bk, bool prio,
              c10::optional<torch::Tensor> dict_sizes) {
  CHECK_IN(xT); CHECK_IN(WT); CHECK_IN(bias); CHECK_IN(c_out);
  CHECK_IN(loss_parts); CHECK_IN(fired);
  int M = WT.size(0), d = WT.size(1), n = WT.size(2);
  int B = xT.size(1);
  TORCH_CHECK(B % 4 == 0 && n % 4 == 0, "B and n must be multiples of 4");
  const int* ds_p = nullptr;
  if (dict_sizes.has_value()) {
    TORCH_CHECK(dict_sizes->is_cuda() && dict_sizes->is_contiguous() &&
                dict_sizes->scalar_type() == torch::kInt32, "dict_sizes must be int32 GPU");
    ds_p = dict_sizes->data_ptr<int>();
  }
  dim3 grid(cdiv(n, BN), cdiv(B, BM), M);
  if (bk == 16)
    hipLaunchKernelGGL((k_enc_fwd2_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       xT.data_ptr<float>(), WT.data_ptr<float>(),
                       bias.data_ptr<float>(), c_out.data_ptr<float>(),
                       loss_parts.data_ptr<float>(), fired.data_ptr<float>(),
                       B, d, n, (int)mode, prio ? 1 : 0, ds_p);
  else
    hipLaunchKernelGGL((k_enc_fwd2_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       xT.data_ptr<float>(), WT.data_ptr<float>(),
                       bias.data_ptr<float>(), c_out.data_ptr<float>(),
                       loss_parts.data_ptr<float>(), fired.data_ptr<float>(),
                       B, d, n, (int)mode, prio ? 1 : 0, ds_p);
}

void gc2(torch::Tensor rT, torch::Tensor WT, torch::Tensor c,
         torch::Tensor l1_alpha, torch::Tensor gpre, torch::Tensor g_bias,
         int64_t bk, bool prio, int64_t gc_mode) {
  CHECK_IN(rT); CHECK_IN(WT); CHECK_IN(c); CHECK_IN(l1_alpha);
  CHECK_IN(gpre); CHECK_IN(g_bias);
  int M = WT.size(0), d = WT.size(1), n = WT.size(2);
  int B = rT.size(2);
  dim3 grid(cdiv(n, BN), cdiv(B, BM), M);
  if (bk == 16)
    hipLaunchKernelGGL((k_gc2_t<16, 6>), grid, dim3(NTHREADS), 0, cur_stream(),
                       rT.data_ptr<float>(), WT.data_ptr<float>(),
                       c.data_ptr<float>(), l1_alpha.data_ptr<float>(),
                       gpre.data_ptr<float>(), g_bias.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0, (int)gc_mode);
  else
    hipLaunchKernelGGL((k_gc2_t<32, 4>), grid, dim3(NTHREADS), 0, cur_stream(),
                       rT.data_ptr<float>(), WT.data_ptr<float>(),
                       c.data_ptr<float>(), l1_alpha.data_ptr<float>(),
                       gpre.data_ptr<float>(), g_bias.data_ptr<float>(),
                       B, d, n, prio ? 1 : 0, (int)gc_mode);
}

void topk_select(torch::Tensor scores, torch::Tensor c_out, torch::Tensor fired,
                 torch::Tensor ks) {
  CHECK_IN(scores); CHECK_IN(c_out); CHECK_IN(fired);
  TORCH_CHECK(ks.is_cuda() && ks.is_contiguous() && ks.scalar_type() == torch::kInt32,
              "ks must be int32 GPU");
  int M = scores.size(0), B = scores.size(1), n = scores.size(2);
  dim3 grid(B, 1, M);
  hipLaunchKernelGGL(k_topk_select, grid, dim3(TOPK_T), 0, cur_stream(),
                     scores.data_ptr<float>(), c_out.data_ptr<float>(),
                     fired.data_ptr<float>(), ks.data_ptr<int>(), B, n);
}

void resample(torch::Tensor fired, torch::Tensor new_rows, torch::Tensor enc_scale,
              torch::Tensor W, torch::Tensor mu_w, torch::Tensor nu_w,
              c10::optional<torch::Tensor> dec,
              c10::optional<torch::Tensor> mu_d, c10::optional<torch::Tensor> nu_d,
              c10::optional<torch::Tensor> bias,
              c10::optional<torch::Tensor> mu_b, c10::optional<torch::Tensor> nu_b,
              torch::Tensor counts_out) {
  CHECK_IN(fired); CHECK_IN(new_rows); CHECK_IN(enc_scale);
  CHECK_IN(W); CHECK_IN(mu_w); CHECK_IN(nu_w);
  TORCH_CHECK(counts_out.is_cuda() && counts_out.scalar_type() == torch::kInt32,
              "counts_out must be int32 GPU");
  int M = W.size(0), n = W.size(1), d = W.size(2);
  int n_track = new_rows.size(1);
  float* dec_p = nullptr; float* mud_p = nullptr; float* nud_p = nullptr;
  if (dec.has_value()) {
    CHECK_IN(dec.value()); CHECK_IN(mu_d.value()); CHECK_IN(nu_d.value());
    dec_p = dec->data_ptr<float>();
    mud_p = mu_d->data_ptr<float>();
    nud_p = nu_d->data_ptr<float>();
  }
  float* b_p = nullptr; float* mub_p = nullptr; float* nub_p = nullptr;
  if (bias.has_value()) {
    CHECK_IN(bias.value()); CHECK_IN(mu_b.value()); CHECK_IN(nu_b.value());
    b_p = bias->data_ptr<float>();
    mub_p = mu_b->data_ptr<float>();
    nub_p = nu_b->data_ptr<float>();
  }
  hipLaunchKernelGGL(k_resample, dim3(M), dim3(RSMP_T), 0, cur_stream(),
                     fired.data_ptr<float>(), new_rows.data_ptr<float>(),
                     enc_scale.data_ptr<float>(), W.data_ptr<float>(),
                     mu_w.data_ptr<float>(), nu_w.data_ptr<float>(),
                     dec_p, mud_p, nud_p, b_p, mub_p, nub_p,
                     counts_out.data_ptr<int>(), n, d, n_track);
}

void lista_bwd_elem(torch::Tensor g_y, c10::optional<torch::Tensor> carry_in,
                    torch::Tensor r, torch::Tensor theta, torch::Tensor x,
                    torch::Tensor x_prev, torch::Tensor mom,
                    torch::Tensor g_r, torch::Tensor carry_out,
                    torch::Tensor g_theta, torch::Tensor g_rho) {
  CHECK_IN(g_y); CHECK_IN(r); CHECK_IN(theta); CHECK_IN(x); CHECK_IN(x_prev);
  CHECK_IN(mom); CHECK_IN(g_r); CHECK_IN(carry_out); CHECK_IN(g_theta); CHECK_IN(g_rho);
  const float* ci = nullptr;
  if (carry_in.has_value()) {
    CHECK_IN(carry_in.value());
    ci = carry_in->data_ptr<float>();
  }
  int M = g_y.size(0), B = g_y.size(1), n = g_y.size(2);
  dim3 grid(cdiv(n, LBW_COLS), cdiv(B, LBW_ROWS), M);
  hipLaunchKernelGGL(k_lista_bwd_elem, grid, dim3(LBW_COLS), 0, cur_stream(),
                     g_y.data_ptr<float>(), ci, r.data_ptr<float>(),
                     theta.data_ptr<float>(), x.data_ptr<float>(),
                     x_prev.data_ptr<float>(), mom.data_ptr<float>(),
                     g_r.data_ptr<float>(), carry_out.data_ptr<float>(),
                     g_theta.data_ptr<float>(), g_rho.data_ptr<float>(), B, n);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("row_norms", &row_norms, "dictionary row norms + clamped inverses");
  m.def("transpose_scale", &transpose_scale, "batched [R,C]->[C,R] transpose with row scale");
  m.def("enc_fwd2", &enc_fwd2, "enc forward, pre-transposed operands (all-direct staging)",
        py::arg("xT"), py::arg("WT"), py::arg("bias"), py::arg("c_out"),
        py::arg("loss_parts"), py::arg("fired"), py::arg("mode"),
        py::arg("bk") = 32, py::arg("prio") = false,
        py::arg("dict_sizes") = py::none());
  m.def("gc2", &gc2, "code-grad, pre-transposed operands",
        py::arg("rT"), py::arg("WT"), py::arg("c"), py::arg("l1_alpha"),
        py::arg("gpre"), py::arg("g_bias"),
        py::arg("bk") = 32, py::arg("prio") = false, py::arg("gc_mode") = 0);
  m.def("enc_fwd", &enc_fwd, "fused encoder GEMM + bias + ReLU/TopK/gate (+L1, fired)",
        py::arg("x"), py::arg("Wenc"), py::arg("bias"), py::arg("inv_norms"),
        py::arg("c_out"), py::arg("loss_parts"), py::arg("fired"), py::arg("mode"),
        py::arg("bk") = 32, py::arg("prio") = false, py::arg("bn") = 128,
        py::arg("act_scale") = py::none(), py::arg("act_gain") = py::none(),
        py::arg("u_out") = py::none(), py::arg("dict_sizes") = py::none(),
        py::arg("y_in") = py::none(), py::arg("x_prev") = py::none(),
        py::arg("x_out") = py::none(), py::arg("mom") = py::none());
  m.def("topk_select", &topk_select, "per-row radix top-k + scatter (+fired)",
        py::arg("scores"), py::arg("c_out"), py::arg("fired"), py::arg("ks"));
  m.def("resample", &resample, "fused dead-neuron resample (K14): rank dead, rewrite rows, zero Adam state",
        py::arg("fired"), py::arg("new_rows"), py::arg("enc_scale"),
        py::arg("W"), py::arg("mu_w"), py::arg("nu_w"),
        py::arg("dec") = py::none(), py::arg("mu_d") = py::none(), py::arg("nu_d") = py::none(),
        py::arg("bias") = py::none(), py::arg("mu_b") = py::none(), py::arg("nu_b") = py::none(),
        py::arg("counts_out"));
  m.def("lista_bwd_elem", &lista_bwd_elem, "fused LISTA backward elementwise pass",
        py::arg("g_y"), py::arg("carry_in"), py::arg("r"), py::arg("theta"),
        py::arg("x"), py::arg("x_prev"), py::arg("mom"), py::arg("g_r"),
        py::arg("carry_out"), py::arg("g_theta"), py::arg("g_rho"));
  m.def("gc_thresh", &gc_thresh, "code-grad through the threshold gate (+gain/scale grads)",
        py::arg("r"), py::arg("Wdec"), py::arg("inv_norms"), py::arg("c"),
        py::arg("u"), py::arg("act_scale"), py::arg("l1_alpha"), py::arg("gpre"),
        py::arg("g_gain"), py::arg("g_scale"),
        py::arg("bk") = 32, py::arg("prio") = false);
  m.def("dec_fwd", &dec_fwd, "fused decoder GEMM - x (+MSE partial)",
        py::arg("c"), py::arg("Wdec"), py::arg("inv_norms"), py::arg("x"),
        py::arg("r_out"), py::arg("loss_parts"),
        py::arg("bk") = 32, py::arg("prio") = false, py::arg("bn") = 128);
  m.def("gc", &gc, "code-gradient GEMM + relu/reverse mask + l1 term (+bias grad)",
        py::arg("r"), py::arg("Wdec"), py::arg("inv_norms"), py::arg("c"),
        py::arg("l1_alpha"), py::arg("gpre"), py::arg("g_bias"),
        py::arg("bk") = 32, py::arg("prio") = false, py::arg("gc_mode") = 0,
        py::arg("bn") = 128);
  m.def("grad_w", &grad_w, "gw = beta*gw + alpha * P^T Q (batched over M)",
        py::arg("P"), py::arg("Q"), py::arg("gw"), py::arg("alpha"), py::arg("beta"),
        py::arg("bk") = 32, py::arg("prio") = false, py::arg("bn") = 128);
  m.def("project_adam", &project_adam, "renorm-gradient projection + Adam",
        py::arg("W"), py::arg("gw"), py::arg("norms"), py::arg("mu"),
        py::arg("nu"), py::arg("step_no"), py::arg("n_per_model"),
        py::arg("lr"), py::arg("b1"), py::arg("b2"), py::arg("eps_adam"),
        py::arg("eps_norm"), py::arg("project"),
        py::arg("w_used") = py::none(), py::arg("clamp_mask") = false,
        py::arg("lr_mult") = py::none());
  m.def("colsum", &colsum, "out[m,j] = alpha * sum_b X[m,b,j] (coalesced column sum)",
        py::arg("X"), py::arg("out"), py::arg("alpha") = 1.0,
        py::arg("absval") = false);
  m.def("bias_adam", &bias_adam, "Adam on bias with L2-norm decay",
        py::arg("bias"), py::arg("g_bias"), py::arg("decay"), py::arg("mu"),
        py::arg("nu"), py::arg("step_no"), py::arg("lr"), py::arg("b1"),
        py::arg("b2"), py::arg("eps_adam"), py::arg("lr_mult") = py::none());
}
