// torch bindings for the baton_amd gfx950 kernels.
//
// Thin layer: validate tensors (contiguous, dtype, device), pull raw
// pointers and the current HIP stream, call the launchers (launchers.h).
// No compute here — every FLOP is in the .hip kernels.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "launchers.h"

namespace {

bool is_bf16(const at::Tensor& t) { return t.scalar_type() == at::kBFloat16; }

void check_compute(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kFloat || t.scalar_type() == at::kBFloat16,
              name, " must be fp32 or bf16");
}

hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

// NN dgrad routing: below this many B bytes the native NN staging beats a
// transpose-to-NT pass (measured on BERT-base dgrad shapes); overridable
// for A/B runs via BATON_NN_TRANSPOSE_BYTES.
long long nn_transpose_bytes() {
  static long long v = [] {
    const char* e = std::getenv("BATON_NN_TRANSPOSE_BYTES");
    return e ? std::atoll(e) : (long long)(1 << 20);
  }();
  return v;
}

}  // namespace

// ---- optim -----------------------------------------------------------------

void sgd_step(at::Tensor p, at::Tensor g, at::Tensor m, double lr,
              double momentum, double weight_decay, bool zero_grad) {
  check_compute(p, "p");
  check_compute(g, "g");
  TORCH_CHECK(p.scalar_type() == g.scalar_type(), "p/g dtype mismatch");
  TORCH_CHECK(p.numel() == g.numel(), "p/g numel mismatch");
  float* mptr = nullptr;
  if (m.defined() && m.numel() > 0) {
    TORCH_CHECK(m.scalar_type() == at::kFloat && m.is_contiguous());
    TORCH_CHECK(m.numel() == p.numel());
    mptr = m.data_ptr<float>();
  }
  launch_sgd(is_bf16(p), p.data_ptr(), g.data_ptr(), mptr, p.numel(), (float)lr,
             (float)momentum, (float)weight_decay, zero_grad ? 1 : 0,
             stream());
}

void adam_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
               double lr, double beta1, double beta2, double eps,
               double weight_decay, double bc1, double bc2, bool zero_grad) {
  check_compute(p, "p");
  check_compute(g, "g");
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel() &&
              p.numel() == v.numel());
  launch_adam(is_bf16(p), p.data_ptr(), g.data_ptr(), m.data_ptr<float>(),
              v.data_ptr<float>(), p.numel(), (float)lr, (float)beta1,
              (float)beta2, (float)eps, (float)weight_decay,
              (float)(1.0 / bc1), (float)(1.0 / std::sqrt(bc2)),
              zero_grad ? 1 : 0, stream());
}

// ---- fedmath ---------------------------------------------------------------

void scale_cast(at::Tensor dst, at::Tensor src, double alpha) {
  TORCH_CHECK(dst.scalar_type() == at::kFloat && dst.is_contiguous());
  check_compute(src, "src");
  TORCH_CHECK(dst.numel() == src.numel());
  launch_scale_cast(is_bf16(src), dst.data_ptr<float>(), src.data_ptr(),
                    src.numel(), (float)alpha, stream());
}

void cast_copy(at::Tensor dst, at::Tensor src) {
  check_compute(dst, "dst");
  TORCH_CHECK(src.scalar_type() == at::kFloat && src.is_contiguous());
  TORCH_CHECK(dst.numel() == src.numel());
  launch_cast_copy(is_bf16(dst), dst.data_ptr(), src.data_ptr<float>(),
                   dst.numel(), stream());
}

at::Tensor colsum(at::Tensor x) {
  check_compute(x, "x");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto out = at::zeros({C}, x.options().dtype(at::kFloat));
  launch_colsum(is_bf16(x), x.data_ptr(), out.data_ptr<float>(), R, C,
                stream());
  return out;
}

void axpby(at::Tensor y, at::Tensor x, double a, double b) {
  TORCH_CHECK(y.scalar_type() == at::kFloat && x.scalar_type() == at::kFloat);
  TORCH_CHECK(y.numel() == x.numel());
  launch_axpby(y.data_ptr<float>(), x.data_ptr<float>(), y.numel(), (float)a,
               (float)b, stream());
}

// ---- loss ------------------------------------------------------------------

at::Tensor mse_fwd(at::Tensor x, at::Tensor y) {
  check_compute(x, "x");
  check_compute(y, "y");
  TORCH_CHECK(x.numel() == y.numel());
  auto out = at::zeros({}, x.options().dtype(at::kFloat));
  launch_mse_fwd(is_bf16(x), x.data_ptr(), y.data_ptr(), out.data_ptr<float>(),
                 x.numel(), stream());
  launch_scale_scalar(out.data_ptr<float>(), 1.f / (float)x.numel(), stream());
  return out;
}

at::Tensor mse_bwd(at::Tensor x, at::Tensor y, at::Tensor dout) {
  check_compute(x, "x");
  check_compute(y, "y");
  TORCH_CHECK(dout.scalar_type() == at::kFloat);
  auto dx = at::empty_like(x);
  launch_mse_bwd(is_bf16(x), x.data_ptr(), y.data_ptr(), dout.data_ptr<float>(),
                 dx.data_ptr(), x.numel(), stream());
  return dx;
}

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target) {
  check_compute(logits, "logits");
  TORCH_CHECK(logits.dim() == 2);
  TORCH_CHECK(target.scalar_type() == at::kLong && target.is_contiguous());
  int B = logits.size(0), C = logits.size(1);
  TORCH_CHECK(target.numel() == B);
  auto lse = at::empty({B}, logits.options().dtype(at::kFloat));
  auto loss = at::zeros({}, logits.options().dtype(at::kFloat));
  launch_ce_fwd(is_bf16(logits), logits.data_ptr(), reinterpret_cast<const long long*>(target.data_ptr<int64_t>()),
                lse.data_ptr<float>(), loss.data_ptr<float>(), B, C, stream());
  launch_scale_scalar(loss.data_ptr<float>(), 1.f / (float)B, stream());
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor dout) {
  check_compute(logits, "logits");
  int B = logits.size(0), C = logits.size(1);
  auto dx = at::empty_like(logits);
  launch_ce_bwd(is_bf16(logits), logits.data_ptr(), reinterpret_cast<const long long*>(target.data_ptr<int64_t>()),
                lse.data_ptr<float>(), dout.data_ptr<float>(), dx.data_ptr(), B,
                C, stream());
  return dx;
}

// ---- layernorm -------------------------------------------------------------

std::vector<at::Tensor> ln_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                               double eps) {
  check_compute(x, "x");
  check_compute(w, "w");
  check_compute(b, "b");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto y = at::empty_like(x);
  auto mean = at::empty({R}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  launch_ln_fwd(is_bf16(x), x.data_ptr(), nullptr, w.data_ptr(), b.data_ptr(),
                y.data_ptr(), nullptr, mean.data_ptr<float>(),
                rstd.data_ptr<float>(), (int)R, C, (float)eps, stream());
  return {y, mean, rstd};
}

// fused residual-add + LayerNorm: y = LN(x + res); also returns z = x + res
// (saved for backward / reused as the residual stream)
std::vector<at::Tensor> ln_add_fwd(at::Tensor x, at::Tensor res, at::Tensor w,
                                   at::Tensor b, double eps) {
  check_compute(x, "x");
  check_compute(res, "res");
  TORCH_CHECK(x.sizes() == res.sizes() && x.scalar_type() == res.scalar_type());
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto y = at::empty_like(x);
  auto z = at::empty_like(x);
  auto mean = at::empty({R}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  launch_ln_fwd(is_bf16(x), x.data_ptr(), res.data_ptr(), w.data_ptr(),
                b.data_ptr(), y.data_ptr(), z.data_ptr(),
                mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)R, C,
                (float)eps, stream());
  return {y, z, mean, rstd};
}

std::vector<at::Tensor> ln_bwd(at::Tensor x, at::Tensor dy, at::Tensor w,
                               at::Tensor mean, at::Tensor rstd) {
  check_compute(x, "x");
  check_compute(dy, "dy");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({C}, x.options().dtype(at::kFloat));
  auto db = at::zeros({C}, x.options().dtype(at::kFloat));
  launch_ln_bwd_dx(is_bf16(x), x.data_ptr(), dy.data_ptr(), w.data_ptr(),
                   mean.data_ptr<float>(), rstd.data_ptr<float>(), nullptr,
                   dx.data_ptr(), (int)R, C, stream());
  launch_ln_bwd_dwdb(is_bf16(x), x.data_ptr(), dy.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     dw.data_ptr<float>(), db.data_ptr<float>(), (int)R, C,
                     stream());
  return {dx, dw, db};
}

// ---- batchnorm (x viewed as [M, C], i.e. NHWC flattened) -------------------

std::vector<at::Tensor> bn_fwd_train(at::Tensor x, at::Tensor gamma,
                                     at::Tensor beta, at::Tensor running_mean,
                                     at::Tensor running_var, double momentum,
                                     double eps, bool relu,
                                     c10::optional<at::Tensor> res_opt) {
  at::Tensor res = res_opt.value_or(at::Tensor());
  const bool has_res = res.defined() && res.numel() > 0;
  if (has_res) {
    check_compute(res, "res");
    TORCH_CHECK(res.sizes() == x.sizes() && res.scalar_type() == x.scalar_type(),
                "bn res must match x");
    TORCH_CHECK(relu, "fused residual add implies fused relu");
  }
  check_compute(x, "x");
  int C = x.size(-1);
  long long M = x.numel() / C;
  TORCH_CHECK(gamma.scalar_type() == at::kFloat, "bn gamma must be fp32");
  auto sum = at::zeros({C}, x.options().dtype(at::kFloat));
  auto sumsq = at::zeros({C}, x.options().dtype(at::kFloat));
  auto mean = at::empty({C}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({C}, x.options().dtype(at::kFloat));
  auto y = at::empty_like(x);
  launch_bn_stats(is_bf16(x), x.data_ptr(), sum.data_ptr<float>(),
                  sumsq.data_ptr<float>(), M, C, stream());
  float* rm = running_mean.defined() && running_mean.numel() > 0
                  ? running_mean.data_ptr<float>()
                  : nullptr;
  float* rv = rm ? running_var.data_ptr<float>() : nullptr;
  launch_bn_finalize(sum.data_ptr<float>(), sumsq.data_ptr<float>(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), rm, rv, M,
                     C, (float)eps, (float)momentum, stream());
  launch_bn_norm(is_bf16(x), relu, x.data_ptr(),
                 has_res ? res.data_ptr() : nullptr, mean.data_ptr<float>(),
                 rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                 beta.data_ptr<float>(), y.data_ptr(), M, C, stream());
  return {y, mean, rstd};
}

at::Tensor bn_fwd_eval(at::Tensor x, at::Tensor gamma, at::Tensor beta,
                       at::Tensor mean, at::Tensor rstd, bool relu) {
  check_compute(x, "x");
  int C = x.size(-1);
  long long M = x.numel() / C;
  auto y = at::empty_like(x);
  launch_bn_norm(is_bf16(x), relu, x.data_ptr(), nullptr,
                 mean.data_ptr<float>(), rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                 beta.data_ptr<float>(), y.data_ptr(), M, C, stream());
  return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor y_post,
                               at::Tensor mean, at::Tensor rstd,
                               at::Tensor gamma, bool relu, bool want_dres) {
  check_compute(x, "x");
  check_compute(dy, "dy");
  int C = x.size(-1);
  long long M = x.numel() / C;
  auto sum_dy = at::zeros({C}, x.options().dtype(at::kFloat));
  auto sum_dyx = at::zeros({C}, x.options().dtype(at::kFloat));
  auto dx = at::empty_like(x);
  at::Tensor dres;
  void* dres_ptr = nullptr;
  if (want_dres) {
    // fused y = relu(bn(x) + res) backward: dres = relu-masked dy, written
    // by the same dx pass. Needs the vectorized path (resnet channels).
    const int velems = is_bf16(x) ? 8 : 4;
    TORCH_CHECK(relu && (C % velems) == 0 && C <= 4096,
                "bn dres needs the fused-relu vectorized path");
    dres = at::empty_like(x);
    dres_ptr = dres.data_ptr();
  }
  const void* ypost_ptr = relu ? y_post.data_ptr() : x.data_ptr();
  launch_bn_bwd_stats(is_bf16(x), relu, x.data_ptr(), dy.data_ptr(), ypost_ptr,
                      mean.data_ptr<float>(), rstd.data_ptr<float>(),
                      sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(), M, C,
                      stream());
  launch_bn_bwd_dx(is_bf16(x), relu, x.data_ptr(), dy.data_ptr(), ypost_ptr,
                   mean.data_ptr<float>(), rstd.data_ptr<float>(),
                   gamma.data_ptr<float>(), sum_dy.data_ptr<float>(),
                   sum_dyx.data_ptr<float>(), dx.data_ptr(), dres_ptr, M, C,
                   stream());
  // dgamma = sum_dyx, dbeta = sum_dy (fp32)
  if (want_dres) return {dx, sum_dyx, sum_dy, dres};
  return {dx, sum_dyx, sum_dy};
}

// ---- elementwise -----------------------------------------------------------

at::Tensor relu_fwd(at::Tensor x) {
  check_compute(x, "x");
  auto y = at::empty_like(x);
  launch_relu_fwd(is_bf16(x), x.data_ptr(), y.data_ptr(), x.numel(), stream());
  return y;
}

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y) {
  check_compute(dy, "dy");
  auto dx = at::empty_like(dy);
  launch_relu_bwd(is_bf16(dy), dy.data_ptr(), y.data_ptr(), dx.data_ptr(),
                  dy.numel(), stream());
  return dx;
}

at::Tensor add_relu_fwd(at::Tensor a, at::Tensor b) {
  check_compute(a, "a");
  check_compute(b, "b");
  TORCH_CHECK(a.numel() == b.numel());
  auto y = at::empty_like(a);
  launch_add_relu_fwd(is_bf16(a), a.data_ptr(), b.data_ptr(), y.data_ptr(),
                      a.numel(), stream());
  return y;
}

at::Tensor add_scaled_fwd(at::Tensor a, at::Tensor b, double alpha) {
  check_compute(a, "a");
  check_compute(b, "b");
  TORCH_CHECK(a.numel() == b.numel());
  auto z = at::empty_like(a);
  launch_add_scaled_fwd(is_bf16(a), a.data_ptr(), b.data_ptr(), z.data_ptr(),
                        (float)alpha, a.numel(), stream());
  return z;
}

at::Tensor scale_fwd(at::Tensor x, double alpha) {
  check_compute(x, "x");
  auto z = at::empty_like(x);
  launch_scale_fwd(is_bf16(x), x.data_ptr(), z.data_ptr(), (float)alpha,
                   x.numel(), stream());
  return z;
}

at::Tensor gelu_fwd(at::Tensor x) {
  check_compute(x, "x");
  auto y = at::empty_like(x);
  launch_gelu_fwd(is_bf16(x), x.data_ptr(), y.data_ptr(), x.numel(), stream());
  return y;
}

at::Tensor gelu_bwd(at::Tensor dy, at::Tensor x) {
  check_compute(dy, "dy");
  auto dx = at::empty_like(dy);
  launch_gelu_bwd(is_bf16(dy), dy.data_ptr(), x.data_ptr(), dx.data_ptr(),
                  dy.numel(), stream());
  return dx;
}

// ---- gemm ------------------------------------------------------------------
// layout: 0 = NT (C=A@B^T, fwd), 1 = NN (C=A@B, dgrad), 2 = TN (C=A^T@B, wgrad)

at::Tensor gemm(at::Tensor A, at::Tensor B, int64_t layout,
                c10::optional<at::Tensor> bias_opt, bool relu, bool out_f32,
                double alpha, double beta, c10::optional<at::Tensor> C_opt,
                bool direct = false) {
  at::Tensor bias = bias_opt.value_or(at::Tensor());
  at::Tensor C_in = C_opt.value_or(at::Tensor());
  check_compute(A, "A");
  check_compute(B, "B");
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2, "gemm wants 2-D tensors");
  TORCH_CHECK(A.scalar_type() == B.scalar_type());
  int M, N, K;
  if (layout == 0) {        // A[M,K] @ B[N,K]^T
    M = A.size(0); K = A.size(1); N = B.size(0);
    TORCH_CHECK(B.size(1) == K, "NT shape mismatch");
  } else if (layout == 1) { // A[M,K'] @ B[K',N]
    M = A.size(0); K = A.size(1); N = B.size(1);
    TORCH_CHECK(B.size(0) == K, "NN shape mismatch");
  } else {                  // A[K,M]^T @ B[K,N]
    K = A.size(0); M = A.size(1); N = B.size(1);
    TORCH_CHECK(B.size(0) == K, "TN shape mismatch");
  }
  auto out_dtype = out_f32 ? at::kFloat : A.scalar_type();
  at::Tensor C;
  if (C_in.defined() && C_in.numel() > 0) {
    TORCH_CHECK(C_in.size(0) == M && C_in.size(1) == N);
    C = C_in;
  } else {
    TORCH_CHECK(beta == 0.0, "beta != 0 needs C_in");
    C = at::empty({M, N}, A.options().dtype(out_dtype));
  }
  const float* bias_ptr = nullptr;
  if (bias.defined() && bias.numel() > 0) {
    TORCH_CHECK(bias.scalar_type() == at::kFloat && bias.numel() == N,
                "bias must be fp32 [N]");
    bias_ptr = bias.data_ptr<float>();
  }
  // ---- dispatch (see gemm.hip header): the glds NT path is the fast one,
  // so big transposed operands are re-laid-out once; small-tile long-K
  // cases split the contraction over grid.z with fp32 accumulation.
  // `direct` skips the transpose-to-NT re-layouts (split-K still applies):
  // for a skinny-M NN like the LoRA dB^T (= xa^T @ dz), transposing the
  // big [tokens, out] upstream grad costs more than the NT gain.
  const bool plain = (beta == 0.0 && bias_ptr == nullptr && !relu);
  at::Tensor Au = A, Bu = B;   // operands in NT orientation when routed
  int eff_layout = (int)layout;
  if (!direct && plain && layout == 2 && !out_f32) {
    // TN -> NT: transpose both (any shape; bounds handled by staging)
    auto At = at::empty({M, K}, A.options());
    auto Bt = at::empty({N, K}, B.options());
    launch_transpose(is_bf16(A), A.data_ptr(), At.data_ptr(), K, M, stream());
    launch_transpose(is_bf16(B), B.data_ptr(), Bt.data_ptr(), K, N, stream());
    Au = At; Bu = Bt; eff_layout = 0;
  } else if (!direct && plain && layout == 1 &&
             (long long)B.numel() * B.element_size() >= nn_transpose_bytes()) {
    // big NN: transpose B -> NT
    auto Bt = at::empty({N, K}, B.options());
    launch_transpose(is_bf16(B), B.data_ptr(), Bt.data_ptr(), K, N, stream());
    Bu = Bt; eff_layout = 0;
  }
  // eff_layout 2 reaches here only with `direct` (otherwise TN was
  // re-laid-out to NT above): native-TN split-K, no transposes
  if (plain && alpha == 1.0 && eff_layout <= 2) {
    // split-K when the tile grid underfills the chip and K is long
    int bn_guess = N <= 32 ? 32 : 128;
    long long tiles = ((long long)(M + 127) / 128) * ((N + bn_guess - 1) / bn_guess);
    if (tiles < 256 && K >= 1024) {
      // 8-phase split-K with per-slice slabs + reduce (no fp32 atomics —
      // the atomic 8-phase variant measured slower, r1 profiles). Covers
      // the long-K skinny-tile wgrads (BERT dW: 256-divisible M/N).
      if (eff_layout == 0 && is_bf16(Au) && M % 256 == 0 && N % 256 == 0) {
        long long t256 = ((long long)M / 256) * (N / 256);
        int splitk = 1;
        while ((long long)splitk * 2 * t256 <= 384 && K % (splitk * 2) == 0 &&
               K / (splitk * 2) >= 256)
          splitk *= 2;
        if (splitk > 1 && K % splitk == 0 && (K / splitk) % 32 == 0) {
          auto slabs = at::empty({(long long)splitk * M * N},
                                 A.options().dtype(at::kFloat));
          if (launch_gemm_nt_8ph_splitk(
                  Au.data_ptr(), Bu.data_ptr(), slabs.data_ptr<float>(),
                  C.data_ptr(), out_dtype != at::kFloat, M, N, K, splitk, 1,
                  stream()))
            return C;
        }
      }
      auto C32 = (out_dtype == at::kFloat)
                     ? C.zero_()
                     : at::zeros({M, N}, A.options().dtype(at::kFloat));
      // (an 8-phase split-K variant with ATOMICS was measured slower here:
      // the 256^2 tile quadruples each block's fp32 atomic output volume —
      // see profiles/; the 128^2 2-phase split-K wins for these shapes)
      launch_gemm_splitk(is_bf16(Au), eff_layout, Au.data_ptr(),
                         Bu.data_ptr(), C32.data_ptr<float>(), M, N, K,
                         stream());
      if (out_dtype == at::kFloat) return C32;
      launch_cast_copy(true, C.data_ptr(), C32.data_ptr<float>(), C.numel(),
                       stream());
      return C;
    }
  }
  launch_gemm(is_bf16(Au), out_f32, eff_layout, relu, Au.data_ptr(),
              Bu.data_ptr(), C.data_ptr(), bias_ptr, M, N, K, (float)alpha,
              (float)beta, stream());
  return C;
}

// ---- conv ------------------------------------------------------------------
// x: [N,H,W,Cin] NHWC contiguous; w: [Cout,KH,KW,Cin]; y: [N,HO,WO,Cout]

at::Tensor conv_fwd(at::Tensor x, at::Tensor w, int64_t stride, int64_t pad) {
  check_compute(x, "x");
  check_compute(w, "w");
  TORCH_CHECK(x.dim() == 4 && w.dim() == 4);
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int Cout = w.size(0), KH = w.size(1), KW = w.size(2);
  TORCH_CHECK(w.size(3) == Cin, "conv channel mismatch");
  int HO = (H + 2 * (int)pad - KH) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - KW) / (int)stride + 1;
  auto y = at::empty({N, HO, WO, Cout}, x.options());
  if (is_bf16(x) && KH == 3 && KW == 3 && stride == 1 && pad == 1 &&
      (long long)H * W % 128 == 0 && 128 % W == 0 && Cin % 32 == 0 &&
      Cout % 64 == 0) {
    // fragment-ordered weight image: one wave B-load = contiguous 1 KiB
    auto wf = w.view({Cout / 16, 16, 9, Cin / 32, 4, 8})
                  .permute({2, 3, 0, 4, 1, 5}).contiguous();
    if (launch_conv_halo_fwd(x.data_ptr(), wf.data_ptr(), y.data_ptr(), N, H,
                             W, Cin, Cout, stream()))
      return y;
  }
  if (is_bf16(x) &&
      launch_conv_fwd_8ph(x.data_ptr(), w.data_ptr(), y.data_ptr(), N, H, W,
                          Cin, Cout, KH, KW, (int)stride, (int)pad, stream()))
    return y;
  launch_conv_fwd(is_bf16(x), x.data_ptr(), w.data_ptr(), y.data_ptr(), N, H, W,
                  Cin, Cout, KH, KW, (int)stride, (int)pad, stream());
  return y;
}

at::Tensor conv_dgrad(at::Tensor dy, at::Tensor w, int64_t H, int64_t W,
                      int64_t stride, int64_t pad) {
  check_compute(dy, "dy");
  check_compute(w, "w");
  int N = dy.size(0), Cout = w.size(0), KH = w.size(1), KW = w.size(2),
      Cin = w.size(3);
  // transposed weight copy [Cin,KH,KW,Cout]: the dgrad B operand becomes a
  // plain k-contiguous row-major matrix (glds-stageable); the permute of a
  // <= few-MB tensor is microseconds
  auto wt = w.permute({3, 1, 2, 0}).contiguous();
  auto dx = at::empty({N, H, W, Cin}, dy.options());
  if (is_bf16(dy) && KH == 3 && KW == 3 && stride == 1 && pad == 1 &&
      (long long)H * W % 128 == 0 && 128 % W == 0 && Cout % 32 == 0 &&
      Cin % 64 == 0) {
    auto wf = wt.view({Cin / 16, 16, 9, Cout / 32, 4, 8})
                  .permute({2, 3, 0, 4, 1, 5}).contiguous();
    if (launch_conv_halo_dgrad(dy.data_ptr(), wf.data_ptr(), dx.data_ptr(), N,
                               (int)H, (int)W, Cin, Cout, stream()))
      return dx;
  }
  if (is_bf16(dy) &&
      launch_conv_dgrad_8ph(dy.data_ptr(), wt.data_ptr(), dx.data_ptr(), N,
                            (int)H, (int)W, Cin, Cout, KH, KW, (int)stride,
                            (int)pad, stream()))
    return dx;
  launch_conv_dgrad(is_bf16(dy), dy.data_ptr(), wt.data_ptr(), dx.data_ptr(), N,
                    (int)H, (int)W, Cin, Cout, KH, KW, (int)stride, (int)pad,
                    stream());
  return dx;
}

at::Tensor conv_wgrad(at::Tensor dy, at::Tensor x, int64_t KH, int64_t KW,
                      int64_t stride, int64_t pad, bool out_f32) {
  check_compute(dy, "dy");
  check_compute(x, "x");
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int Cout = dy.size(3);
  // split-K accumulates in fp32 (zero-init); cast down only if requested.
  // dy transposed once ([P,Cout] -> [Cout,P]) so the wgrad A operand
  // stages direct/glds (k = pixel contiguous).
  long long P = (long long)dy.size(0) * dy.size(1) * dy.size(2);
  auto dyt = at::empty({(long long)Cout, P}, dy.options());
  launch_transpose(is_bf16(dy), dy.data_ptr(), dyt.data_ptr(), P, Cout,
                   stream());
  auto dw32 = at::zeros({Cout, KH, KW, Cin}, x.options().dtype(at::kFloat));
  if (KH == 1 && KW == 1 && stride == 1 && pad == 0) {
    // 1x1 stride-1 wgrad IS a dense GEMM: dw[Cout,Cin] = dy_t[Cout,P] @
    // x[P,Cin], NN, K = pixels — the split-K GEMM runs it at GEMM speed
    // (the conv-path x gather measured ~68 TF on these shapes, the NN
    // split-K ~250 TF). When Cout/Cin are 256-divisible, transpose x once
    // (P is already transposed for dy) and run the 8-phase split-K slab
    // kernel instead (r2: the rn50 bottleneck 1x1 wgrads were 12% of the
    // step on the 2-phase atomic path).
    bool done8 = false;
    if (is_bf16(x) && Cout % 256 == 0 && Cin % 256 == 0 && P % 32 == 0) {
      long long t256 = ((long long)Cout / 256) * (Cin / 256);
      int splitk = 1;
      while ((long long)splitk * 2 * t256 <= 384 && P % (splitk * 2) == 0 &&
             P / (splitk * 2) >= 256)
        splitk *= 2;
      if (splitk > 1 && (P / splitk) % 32 == 0) {
        auto xt = at::empty({(long long)Cin, P}, x.options());
        launch_transpose(true, x.data_ptr(), xt.data_ptr(), P, Cin, stream());
        auto slabs = at::empty({(long long)splitk * Cout * Cin},
                               x.options().dtype(at::kFloat));
        done8 = launch_gemm_nt_8ph_splitk(
            dyt.data_ptr(), xt.data_ptr(), slabs.data_ptr<float>(),
            dw32.data_ptr(), false, Cout, Cin, (int)P, splitk, 1, stream());
      }
    }
    if (!done8)
      launch_gemm_splitk(is_bf16(x), 1, dyt.data_ptr(), x.data_ptr(),
                         dw32.data_ptr<float>(), Cout, Cin, (int)P, stream());
  } else {
    launch_conv_wgrad(is_bf16(x), true, dyt.data_ptr(), x.data_ptr(),
                      dw32.data_ptr(), N, H, W, Cin, Cout, (int)KH, (int)KW,
                      (int)stride, (int)pad, stream());
  }
  if (out_f32 || x.scalar_type() == at::kFloat) return dw32;
  auto dw = at::empty({Cout, KH, KW, Cin}, x.options());
  launch_cast_copy(true, dw.data_ptr(), dw32.data_ptr<float>(), dw.numel(),
                   stream());
  return dw;
}


// Strided batched GEMM over tensor VIEWS: operands may be non-contiguous
// slices of [B, S, h, dh] / [B, S, 3, h, dh] tensors (the attention path
// consumes them in place — no permute copies). Layouts as gemm(); all
// strides in ELEMENTS. z = outer * heads + head; B-operand head index is
// divided by b_group (GQA). C may be a preallocated strided view.
at::Tensor bmm_strided(at::Tensor A, at::Tensor B, c10::optional<at::Tensor> C_in,
                       int64_t layout, int64_t M, int64_t N, int64_t K,
                       int64_t nbatch, int64_t heads, int64_t b_group,
                       double alpha,
                       int64_t saO, int64_t saI, int64_t lda,
                       int64_t sbO, int64_t sbI, int64_t ldb,
                       int64_t scO, int64_t scI, int64_t ldc) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda(), "bmm_strided: GPU tensors required");
  TORCH_CHECK(A.scalar_type() == B.scalar_type());
  TORCH_CHECK(A.scalar_type() == at::kFloat || A.scalar_type() == at::kBFloat16);
  at::Tensor C;
  if (C_in.has_value() && C_in->defined()) {
    C = *C_in;
    TORCH_CHECK(C.is_cuda() && C.scalar_type() == A.scalar_type());
  } else {
    C = at::empty({nbatch, M, N}, A.options());
    scO = (int64_t)heads * M * N;
    scI = (int64_t)M * N;
    ldc = N;
  }
  GemmStrides gs{lda, ldb, ldc, (int)heads, saI, sbI, scI};
  launch_gemm_batched(is_bf16(A), false, (int)layout, false, A.data_ptr(),
                      B.data_ptr(), C.data_ptr(), nullptr, (int)M, (int)N,
                      (int)K, (float)alpha, 0.f, (int)nbatch, saO, sbO, scO,
                      stream(), (int)b_group, gs);
  return C;
}

// ---- batched gemm + softmax (attention) ------------------------------------
// A/B/C are 3-D [nb, *, *]; layout semantics per batch as in gemm().

at::Tensor gemm_batched(at::Tensor A, at::Tensor B, int64_t layout,
                        bool out_f32, double alpha, int64_t b_group) {
  check_compute(A, "A");
  check_compute(B, "B");
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3, "gemm_batched wants 3-D tensors");
  TORCH_CHECK(A.size(0) == B.size(0) * b_group, "batch/group mismatch");
  int nb = A.size(0);
  int M, N, K;
  if (layout == 0) {
    M = A.size(1); K = A.size(2); N = B.size(1);
    TORCH_CHECK(B.size(2) == K);
  } else if (layout == 1) {
    M = A.size(1); K = A.size(2); N = B.size(2);
    TORCH_CHECK(B.size(1) == K);
  } else {
    K = A.size(1); M = A.size(2); N = B.size(2);
    TORCH_CHECK(B.size(1) == K);
  }
  auto C = at::empty({nb, M, N},
                     A.options().dtype(out_f32 ? at::kFloat : A.scalar_type()));
  launch_gemm_batched(is_bf16(A), out_f32, (int)layout, false, A.data_ptr(),
                      B.data_ptr(), C.data_ptr(), nullptr, M, N, K,
                      (float)alpha, 0.f, nb, (long long)A.size(1) * A.size(2),
                      (long long)B.size(1) * B.size(2), (long long)M * N,
                      stream(), (int)b_group);
  return C;
}

at::Tensor softmax_fwd(at::Tensor x, double scale, int64_t causal_seq) {
  check_compute(x, "x");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto y = at::empty_like(x);
  launch_softmax_fwd(is_bf16(x), x.data_ptr(), y.data_ptr(), R, C,
                     (float)scale, (int)causal_seq, stream());
  return y;
}

at::Tensor softmax_bwd(at::Tensor y, at::Tensor dy, double scale) {
  check_compute(y, "y");
  check_compute(dy, "dy");
  int C = y.size(-1);
  long long R = y.numel() / C;
  auto dx = at::empty_like(y);
  launch_softmax_bwd(is_bf16(y), y.data_ptr(), dy.data_ptr(), dx.data_ptr(), R,
                     C, (float)scale, stream());
  return dx;
}


// ---- llama ops -------------------------------------------------------------

std::vector<at::Tensor> rms_fwd(at::Tensor x, at::Tensor w, double eps) {
  check_compute(x, "x");
  check_compute(w, "w");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto y = at::empty_like(x);
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  launch_rms_fwd(is_bf16(x), x.data_ptr(), nullptr, w.data_ptr(), y.data_ptr(),
                 nullptr, rstd.data_ptr<float>(), R, C, (float)eps, stream());
  return {y, rstd};
}

// fused residual-add + RMSNorm: y = RMS(x + res), z = x + res returned for
// the residual stream and backward
std::vector<at::Tensor> rms_add_fwd(at::Tensor x, at::Tensor res, at::Tensor w,
                                    double eps) {
  check_compute(x, "x");
  check_compute(res, "res");
  TORCH_CHECK(x.sizes() == res.sizes() && x.scalar_type() == res.scalar_type());
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto y = at::empty_like(x);
  auto z = at::empty_like(x);
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  launch_rms_fwd(is_bf16(x), x.data_ptr(), res.data_ptr(), w.data_ptr(),
                 y.data_ptr(), z.data_ptr(), rstd.data_ptr<float>(), R, C,
                 (float)eps, stream());
  return {y, z, rstd};
}

std::vector<at::Tensor> rms_bwd(at::Tensor x, at::Tensor dy, at::Tensor w,
                                at::Tensor rstd) {
  check_compute(x, "x");
  check_compute(dy, "dy");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({C}, x.options().dtype(at::kFloat));
  launch_rms_bwd(is_bf16(x), x.data_ptr(), dy.data_ptr(), w.data_ptr(),
                 rstd.data_ptr<float>(), nullptr, dx.data_ptr(),
                 dw.data_ptr<float>(), R, C, stream());
  return {dx, dw};
}

std::vector<at::Tensor> rms_bwd_plus(at::Tensor x, at::Tensor dy, at::Tensor w,
                                     at::Tensor rstd, at::Tensor plus) {
  check_compute(x, "x");
  check_compute(dy, "dy");
  check_compute(plus, "plus");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({C}, x.options().dtype(at::kFloat));
  launch_rms_bwd(is_bf16(x), x.data_ptr(), dy.data_ptr(), w.data_ptr(),
                 rstd.data_ptr<float>(), plus.data_ptr(), dx.data_ptr(),
                 dw.data_ptr<float>(), R, C, stream());
  return {dx, dw};
}

// ln_bwd with the residual-stream gradient fused into dx (dx += plus)
std::vector<at::Tensor> ln_bwd_plus(at::Tensor x, at::Tensor dy, at::Tensor w,
                                    at::Tensor mean, at::Tensor rstd,
                                    at::Tensor plus) {
  check_compute(x, "x");
  check_compute(dy, "dy");
  check_compute(plus, "plus");
  int C = x.size(-1);
  long long R = x.numel() / C;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({C}, x.options().dtype(at::kFloat));
  auto db = at::zeros({C}, x.options().dtype(at::kFloat));
  launch_ln_bwd_dx(is_bf16(x), x.data_ptr(), dy.data_ptr(), w.data_ptr(),
                   mean.data_ptr<float>(), rstd.data_ptr<float>(),
                   plus.data_ptr(), dx.data_ptr(), (int)R, C, stream());
  launch_ln_bwd_dwdb(is_bf16(x), x.data_ptr(), dy.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     dw.data_ptr<float>(), db.data_ptr<float>(), (int)R, C,
                     stream());
  return {dx, dw, db};
}

at::Tensor rope(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                bool inverse) {
  check_compute(x, "x");
  TORCH_CHECK(x.dim() == 4, "rope wants [B,S,H,D]");
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && sin_t.scalar_type() == at::kFloat);
  int S = x.size(1), H = x.size(2), D = x.size(3);
  TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "rope table shape");
  auto y = at::empty_like(x);
  long long total = x.numel() / 2;
  launch_rope(is_bf16(x), inverse, x.data_ptr(), y.data_ptr(),
              cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), total, S, H, D,
              stream());
  return y;
}

at::Tensor silu_mul_fwd(at::Tensor a, at::Tensor b) {
  check_compute(a, "a");
  check_compute(b, "b");
  TORCH_CHECK(a.numel() == b.numel());
  auto y = at::empty_like(a);
  launch_silu_mul_fwd(is_bf16(a), a.data_ptr(), b.data_ptr(), y.data_ptr(),
                      a.numel(), stream());
  return y;
}

std::vector<at::Tensor> silu_mul_bwd(at::Tensor dy, at::Tensor a, at::Tensor b) {
  check_compute(dy, "dy");
  auto da = at::empty_like(a);
  auto db = at::empty_like(b);
  launch_silu_mul_bwd(is_bf16(a), dy.data_ptr(), a.data_ptr(), b.data_ptr(),
                      da.data_ptr(), db.data_ptr(), a.numel(), stream());
  return {da, db};
}

// ---- flash attention -------------------------------------------------------

namespace {
// q/k/v/o are 4-D [B,S,heads,DH] strided VIEWS with contiguous DH
void check_attn_view(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16, name,
              " must be a bf16 GPU tensor");
  TORCH_CHECK(t.dim() == 4, name, " must be [B,S,h,dh]");
  TORCH_CHECK(t.stride(3) == 1, name, " head_dim must be contiguous");
}
void strides3(const at::Tensor& t, long long* out) {
  out[0] = t.stride(0);  // batch
  out[1] = t.stride(1);  // seq
  out[2] = t.stride(2);  // head
}
}  // namespace

std::vector<at::Tensor> flash_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                  at::Tensor o, bool causal) {
  check_attn_view(q, "q");
  check_attn_view(k, "k");
  check_attn_view(v, "v");
  check_attn_view(o, "o");
  const int B = q.size(0), S = q.size(1), H = q.size(2), DH = q.size(3);
  const int KVH = k.size(2);
  TORCH_CHECK(H % KVH == 0, "q heads must be a multiple of kv heads");
  const int G = H / KVH;
  auto lse = at::empty({(long long)B * H, S},
                       q.options().dtype(at::kFloat));
  long long qs[3], ks[3], vs[3], os[3];
  strides3(q, qs); strides3(k, ks); strides3(v, vs); strides3(o, os);
  const float scale = 1.0f / std::sqrt((float)DH);
  TORCH_CHECK(
      launch_flash_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                       lse.data_ptr<float>(), B, S, H, G, DH, scale, causal,
                       qs, ks, vs, os, stream()),
      "flash_fwd: unsupported head_dim ", DH);
  return {o, lse};
}

void flash_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
               at::Tensor dout, at::Tensor lse, at::Tensor dq, at::Tensor dk,
               at::Tensor dv, bool causal) {
  check_attn_view(q, "q");
  check_attn_view(k, "k");
  check_attn_view(v, "v");
  check_attn_view(o, "o");
  check_attn_view(dout, "dout");
  check_attn_view(dq, "dq");
  check_attn_view(dk, "dk");
  check_attn_view(dv, "dv");
  const int B = q.size(0), S = q.size(1), H = q.size(2), DH = q.size(3);
  const int KVH = k.size(2);
  const int G = H / KVH;
  TORCH_CHECK(dk.size(2) == H && dv.size(2) == H,
              "dk/dv are per-q-head (GQA callers group-sum)");
  TORCH_CHECK(lse.scalar_type() == at::kFloat && lse.is_contiguous());
  auto delta = at::empty_like(lse);
  long long qs[3], ks[3], vs[3], os[3], dos[3], dqs[3], dks[3], dvs[3];
  strides3(q, qs); strides3(k, ks); strides3(v, vs); strides3(o, os);
  strides3(dout, dos); strides3(dq, dqs); strides3(dk, dks); strides3(dv, dvs);
  const float scale = 1.0f / std::sqrt((float)DH);
  launch_flash_delta(dout.data_ptr(), o.data_ptr(), delta.data_ptr<float>(),
                     B, S, H, DH, dos, os, stream());
  TORCH_CHECK(
      launch_flash_bwd_dq(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          dout.data_ptr(), lse.data_ptr<float>(),
                          delta.data_ptr<float>(), dq.data_ptr(), B, S, H, G,
                          DH, scale, causal, qs, ks, vs, dos, dqs, stream()),
      "flash_bwd_dq: unsupported head_dim ", DH);
  TORCH_CHECK(
      launch_flash_bwd_dkv(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                           dout.data_ptr(), lse.data_ptr<float>(),
                           delta.data_ptr<float>(), dk.data_ptr(),
                           dv.data_ptr(), B, S, H, G, DH, scale, causal, qs,
                           ks, vs, dos, dks, dvs, stream()),
      "flash_bwd_dkv: unsupported head_dim ", DH);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("flash_fwd", &flash_fwd);
  m.def("flash_bwd", &flash_bwd);
  m.def("rms_add_fwd", &rms_add_fwd);
  m.def("rms_bwd_plus", &rms_bwd_plus);
  m.def("sgd_step", &sgd_step, "fused SGD step", py::arg("p"), py::arg("g"),
        py::arg("m"), py::arg("lr"), py::arg("momentum"),
        py::arg("weight_decay"), py::arg("zero_grad") = false);
  m.def("adam_step", &adam_step, "fused Adam step", py::arg("p"), py::arg("g"),
        py::arg("m"), py::arg("v"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("weight_decay"),
        py::arg("bc1"), py::arg("bc2"), py::arg("zero_grad") = false);
  m.def("scale_cast", &scale_cast);
  m.def("cast_copy", &cast_copy);
  m.def("axpby", &axpby);
  m.def("bmm_strided", &bmm_strided, py::arg("A"), py::arg("B"),
        py::arg("C_in") = py::none(), py::arg("layout") = 0,
        py::arg("M") = 0, py::arg("N") = 0, py::arg("K") = 0,
        py::arg("nbatch") = 1, py::arg("heads") = 1,
        py::arg("b_group") = 1, py::arg("alpha") = 1.0,
        py::arg("saO") = 0, py::arg("saI") = 0, py::arg("lda") = 0,
        py::arg("sbO") = 0, py::arg("sbI") = 0, py::arg("ldb") = 0,
        py::arg("scO") = 0, py::arg("scI") = 0, py::arg("ldc") = 0);
  m.def("colsum", &colsum);
  m.def("mse_fwd", &mse_fwd);
  m.def("mse_bwd", &mse_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("ln_fwd", &ln_fwd);
  m.def("ln_bwd", &ln_bwd);
  m.def("ln_add_fwd", &ln_add_fwd);
  m.def("ln_bwd_plus", &ln_bwd_plus);
  m.def("bn_fwd_train", &bn_fwd_train, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("running_mean"), py::arg("running_var"),
        py::arg("momentum"), py::arg("eps"), py::arg("relu"),
        py::arg("res") = py::none());
  m.def("bn_fwd_eval", &bn_fwd_eval);
  m.def("bn_bwd", &bn_bwd, py::arg("x"), py::arg("dy"), py::arg("y_post"),
        py::arg("mean"), py::arg("rstd"), py::arg("gamma"), py::arg("relu"),
        py::arg("want_dres") = false);
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("add_relu_fwd", &add_relu_fwd);
  m.def("add_scaled_fwd", &add_scaled_fwd);
  m.def("scale_fwd", &scale_fwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("gemm", &gemm, py::arg("A"), py::arg("B"), py::arg("layout"),
        py::arg("bias") = py::none(), py::arg("relu") = false,
        py::arg("out_f32") = false, py::arg("alpha") = 1.0,
        py::arg("beta") = 0.0, py::arg("C_in") = py::none(),
        py::arg("direct") = false);
  m.def("gemm_batched", &gemm_batched, py::arg("A"), py::arg("B"),
        py::arg("layout"), py::arg("out_f32") = false, py::arg("alpha") = 1.0,
        py::arg("b_group") = 1);
  m.def("softmax_fwd", &softmax_fwd, py::arg("x"), py::arg("scale") = 1.0,
        py::arg("causal_seq") = 0);
  m.def("softmax_bwd", &softmax_bwd);
  m.def("rms_fwd", &rms_fwd);
  m.def("rms_bwd", &rms_bwd);
  m.def("rope", &rope, py::arg("x"), py::arg("cos_t"), py::arg("sin_t"),
        py::arg("inverse") = false);
  m.def("silu_mul_fwd", &silu_mul_fwd);
  m.def("silu_mul_bwd", &silu_mul_bwd);
  m.def("conv_fwd", &conv_fwd);
  m.def("conv_dgrad", &conv_dgrad);
  m.def("conv_wgrad", &conv_wgrad);
}
