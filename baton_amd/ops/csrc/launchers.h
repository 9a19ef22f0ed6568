// Host-side launcher declarations: the boundary between the torch binding
// layer (bindings.cpp) and the HIP kernel TUs. Raw pointers + hipStream_t
// only — keeps torch headers out of kernel compiles.
#pragma once
#include <hip/hip_runtime.h>

#include "gemm_strides.h"

// optim.hip
void launch_sgd(bool is_bf16, void* p, void* g, float* m, long long n,
                float lr, float momentum, float wd, int zero_after,
                hipStream_t s);
void launch_adam(bool is_bf16, void* p, void* g, float* m, float* v,
                 long long n, float lr, float b1, float b2, float eps, float wd,
                 float inv_bc1, float inv_sqrt_bc2, int zero_after,
                 hipStream_t s);

// fedmath.hip
void launch_scale_cast(bool src_bf16, float* dst, const void* src, long long n,
                       float alpha, hipStream_t s);
void launch_cast_copy(bool dst_bf16, void* dst, const float* src, long long n,
                      hipStream_t s);
void launch_axpby(float* y, const float* x, long long n, float a, float b,
                  hipStream_t s);

// loss.hip
void launch_mse_fwd(bool is_bf16, const void* x, const void* y, float* out,
                    long long n, hipStream_t s);
void launch_scale_scalar(float* out, float scale, hipStream_t s);
void launch_mse_bwd(bool is_bf16, const void* x, const void* y,
                    const float* dout, void* dx, long long n, hipStream_t s);
void launch_ce_fwd(bool is_bf16, const void* logits, const long long* target,
                   float* lse, float* loss_sum, int B, int C, hipStream_t s);
void launch_ce_bwd(bool is_bf16, const void* logits, const long long* target,
                   const float* lse, const float* dout, void* dx, int B, int C,
                   hipStream_t s);

// norm.hip
void launch_ln_fwd(bool is_bf16, const void* x, const void* res, const void* w,
                   const void* b, void* y, void* zout, float* mean,
                   float* rstd, int R, int C, float eps, hipStream_t s);
void launch_ln_bwd_dx(bool is_bf16, const void* x, const void* dy, const void* w,
                      const float* mean, const float* rstd, const void* plus,
                      void* dx, int R, int C, hipStream_t s);
void launch_ln_bwd_dwdb(bool is_bf16, const void* x, const void* dy,
                        const float* mean, const float* rstd, float* dw,
                        float* db, int R, int C, hipStream_t s);
void launch_bn_stats(bool is_bf16, const void* x, float* sum, float* sumsq,
                     long long M, int C, hipStream_t s);
void launch_bn_finalize(const float* sum, const float* sumsq, float* mean,
                        float* rstd, float* running_mean, float* running_var,
                        long long M, int C, float eps, float momentum,
                        hipStream_t s);
void launch_bn_norm(bool is_bf16, bool relu, const void* x, const void* res,
                    const float* mean,
                    const float* rstd, const float* gamma, const float* beta,
                    void* y, long long M, int C, hipStream_t s);
void launch_bn_bwd_stats(bool is_bf16, bool relu, const void* x, const void* dy,
                         const void* y_post, const float* mean,
                         const float* rstd, float* sum_dy, float* sum_dyx,
                         long long M, int C, hipStream_t s);
void launch_bn_bwd_dx(bool is_bf16, bool relu, const void* x, const void* dy,
                      const void* y_post, const float* mean, const float* rstd,
                      const float* gamma, const float* sum_dy,
                      const float* sum_dyx, void* dx, void* dres, long long M,
                      int C, hipStream_t s);

// elementwise.hip
void launch_relu_fwd(bool is_bf16, const void* x, void* y, long long n,
                     hipStream_t s);
void launch_relu_bwd(bool is_bf16, const void* dy, const void* y, void* dx,
                     long long n, hipStream_t s);
void launch_add_relu_fwd(bool is_bf16, const void* a, const void* b, void* y,
                         long long n, hipStream_t s);
void launch_add_scaled_fwd(bool is_bf16, const void* a, const void* b, void* z,
                           float alpha, long long n, hipStream_t s);
void launch_scale_fwd(bool is_bf16, const void* x, void* z, float alpha,
                      long long n, hipStream_t s);
void launch_gelu_fwd(bool is_bf16, const void* x, void* y, long long n,
                     hipStream_t s);
void launch_gelu_bwd(bool is_bf16, const void* dy, const void* x, void* dx,
                     long long n, hipStream_t s);

// gemm.hip — layout: 0 = NT (fwd), 1 = NN (dgrad), 2 = TN (wgrad)
void launch_gemm(bool in_bf16, bool out_f32, int layout, bool relu,
                 const void* A, const void* B, void* C, const float* bias,
                 int M, int N, int K, float alpha, float beta, hipStream_t s);

// conv.hip — NHWC implicit GEMM
void launch_conv_fwd(bool is_bf16, const void* x, const void* w, void* y,
                     int N, int H, int W, int Cin, int Cout, int KH, int KW,
                     int stride, int pad, hipStream_t s);
void launch_conv_dgrad(bool is_bf16, const void* dy, const void* w, void* dx,
                       int N, int H, int W, int Cin, int Cout, int KH, int KW,
                       int stride, int pad, hipStream_t s);
void launch_conv_wgrad(bool is_bf16, bool out_f32, const void* dy,
                       const void* x, void* dw, int N, int H, int W, int Cin,
                       int Cout, int KH, int KW, int stride, int pad,
                       hipStream_t s);

// gemm.hip batched (attention): grid.z = nbatch, per-batch element strides
void launch_gemm_batched(bool in_bf16, bool out_f32, int layout, bool relu,
                         const void* A, const void* B, void* C,
                         const float* bias, int M, int N, int K, float alpha,
                         float beta, int nbatch, long long strideA,
                         long long strideB, long long strideC, hipStream_t s,
                         int b_group = 1,
                         GemmStrides gs = GemmStrides{0, 0, 0, 1, 0, 0, 0});

// attention.hip — row softmax with scale + optional causal mask
void launch_softmax_fwd(bool is_bf16, const void* x, void* y, long long R,
                        int C, float scale, int causal_seq, hipStream_t s);
void launch_softmax_bwd(bool is_bf16, const void* y, const void* dy, void* dx,
                        long long R, int C, float scale, hipStream_t s);

// llama_ops.hip — RMSNorm / RoPE / SwiGLU
void launch_rms_fwd(bool is_bf16, const void* x, const void* res,
                    const void* w, void* y, void* zout, float* rstd,
                    long long R, int C, float eps, hipStream_t s);
void launch_rms_bwd(bool is_bf16, const void* x, const void* dy, const void* w,
                    const float* rstd, const void* plus, void* dx, float* dw,
                    long long R, int C, hipStream_t s);
void launch_rope(bool is_bf16, bool inverse, const void* x, void* y,
                 const float* cos_t, const float* sin_t, long long total_pairs,
                 int S, int H, int D, hipStream_t s);
void launch_silu_mul_fwd(bool is_bf16, const void* a, const void* b, void* y,
                         long long n, hipStream_t s);
void launch_silu_mul_bwd(bool is_bf16, const void* dy, const void* a,
                         const void* b, void* da, void* db, long long n,
                         hipStream_t s);

// gemm.hip split-K (layouts NT/NN): fp32 atomic accumulation over K slices
void launch_gemm_splitk(bool in_bf16, int layout, const void* A,
                        const void* B, float* C, int M, int N, int K,
                        hipStream_t s);

// fedmath.hip — LDS-tiled matrix transpose [R,C] -> [C,R]
void launch_transpose(bool is_bf16, const void* in, void* out, int R, int C,
                      hipStream_t s);

// fedmath.hip — column sum (bias gradients): out[c] = sum_r x[r,c], fp32
void launch_colsum(bool is_bf16, const void* x, float* out, long long R, int C,
                   hipStream_t s);

// gemm8.hip — deep-pipelined 256x256 8-wave NT kernel (bf16 full tiles);
// returns false when the shape is ineligible (caller falls back)
bool launch_gemm_nt_8ph(const void* A, const void* B, void* C,
                        const float* bias, int M, int N, int K, float alpha,
                        int use_swz, hipStream_t s);

// gemm8.hip — split-K 8-phase: per-slice fp32 slabs + reduce (no atomics)
bool launch_gemm_nt_8ph_splitk(const void* A, const void* B, float* slabs,
                               void* out, bool out_bf16, int M, int N, int K,
                               int splitk, int use_swz, hipStream_t s);

// flash.hip — fused flash attention (bf16, DH in {32,64,128}); strides in
// elements as [batch, seq, head] triples; returns false if DH unsupported
bool launch_flash_fwd(const void* Q, const void* K, const void* V, void* O,
                      float* LSE, int B, int S, int H, int G, int DH,
                      float scale, bool causal,
                      const long long* qs, const long long* ks,
                      const long long* vs, const long long* os,
                      hipStream_t s);
void launch_flash_delta(const void* dO, const void* O, float* delta,
                        int B, int S, int H, int DH,
                        const long long* dos, const long long* os,
                        hipStream_t s);
bool launch_flash_bwd_dq(const void* Q, const void* K, const void* V,
                         const void* dO, const float* LSE, const float* DELTA,
                         void* dQ, int B, int S, int H, int G, int DH,
                         float scale, bool causal,
                         const long long* qs, const long long* ks,
                         const long long* vs, const long long* dos,
                         const long long* dqs, hipStream_t s);
bool launch_flash_bwd_dkv(const void* Q, const void* K, const void* V,
                          const void* dO, const float* LSE, const float* DELTA,
                          void* dK, void* dV, int B, int S, int H, int G,
                          int DH, float scale, bool causal,
                          const long long* qs, const long long* ks,
                          const long long* vs, const long long* dos,
                          const long long* dks, const long long* dvs,
                          hipStream_t s);

// conv8.hip — deep-pipelined 256x256 8-wave conv fwd/dgrad (bf16);
// returns false when the shape is ineligible (caller falls back)
bool launch_conv_fwd_8ph(const void* x, const void* w, void* y, int N, int H,
                         int W, int Cin, int Cout, int KH, int KW, int stride,
                         int pad, hipStream_t s);
bool launch_conv_dgrad_8ph(const void* dy, const void* w_t, void* dx, int N,
                           int H, int W, int Cin, int Cout, int KH, int KW,
                           int stride, int pad, hipStream_t s);

// conv_halo.hip — shared-halo 3x3 stride-1 conv fwd/dgrad for small-channel
// layers whose 128-px tiles stay within one image; false when ineligible
bool launch_conv_halo_fwd(const void* x, const void* w, void* y, int N, int H,
                          int W, int Cin, int Cout, hipStream_t s);
bool launch_conv_halo_dgrad(const void* dy, const void* w_t, void* dx, int N,
                            int H, int W, int Cin, int Cout, hipStream_t s);
