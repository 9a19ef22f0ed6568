// Llama-family kernels (gfx950): RMSNorm, RoPE, SwiGLU — fwd + bwd.
//
// BASELINE.json config 4 (Llama-3-8B LoRA federated fine-tune). All three
// are memory-bound streaming ops: vectorized loads where layout permits,
// fp32 math, one block per row for the norm, grid-stride elsewhere. RoPE
// uses host-precomputed cos/sin tables (guide Appendix B: on-device trig
// turns a memory-bound op VALU-bound).
#include "common.h"

// ---- RMSNorm ---------------------------------------------------------------
// y = x * rstd * w,  rstd = (mean(x^2) + eps)^-1/2;  saves rstd for bwd.
// ADD variant fuses the transformer residual add (guide: fuse elementwise
// work into the producing pass): z = x + res is written ONCE here and the
// norm consumes it — the separate add kernel and one full tensor read
// disappear (the at::add glue in the r1 profiles, VERDICT weak #4).

template <typename T, bool ADD>
__global__ void rms_fwd_kernel(const T* __restrict__ x,
                               const T* __restrict__ res,
                               const T* __restrict__ w, T* __restrict__ y,
                               T* __restrict__ zout,
                               float* __restrict__ rstd_out,
                               long long R, int C, float eps) {
  __shared__ float scratch[kBlock / kWave];
  for (long long r = blockIdx.x; r < R; r += gridDim.x) {
    const T* row = x + r * C;
    T* yrow = y + r * C;
    constexpr int V = VecTraits<T>::kElems;
    const bool vec = (C % V) == 0;      // wave-uniform
    float ss = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V];
        vload16(row + cv * V, f);
        if (ADD) {
          float fr[V];
          vload16(res + r * C + cv * V, fr);
#pragma unroll
          for (int q = 0; q < V; ++q) f[q] += fr[q];
          vstore16(zout + r * C + cv * V, f);
        }
#pragma unroll
        for (int q = 0; q < V; ++q) ss = fmaf(f[q], f[q], ss);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float v = (float)row[c];
        if (ADD) {
          v += (float)res[r * C + c];
          zout[r * C + c] = (T)v;
        }
        ss = fmaf(v, v, ss);
      }
    }
    float sumsq = block_reduce_sum(ss, scratch);
    float rstd = rsqrtf(sumsq / C + eps);
    if (threadIdx.x == 0) rstd_out[r] = rstd;
    // normalize pass re-reads the summed row (same CU wrote it; the
    // inter-CU L1 staleness hazard does not apply to a block's own lines)
    const T* zrow = ADD ? zout + r * C : row;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V], fw[V];
        vload16(zrow + cv * V, f);
        vload16(w + cv * V, fw);
#pragma unroll
        for (int q = 0; q < V; ++q) f[q] = f[q] * rstd * fw[q];
        vstore16(yrow + cv * V, f);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x)
        yrow[c] = (T)((float)zrow[c] * rstd * (float)w[c]);
    }
  }
}

// dx = rstd * (dy*w - xhat * mean(dy*w*xhat)), xhat = x*rstd
// PLUS fuses the residual-stream gradient add (backward of the fused
// z = x + res forward): dx += plus, saving the separate add pass.
template <typename T, bool PLUS>
__global__ void rms_bwd_dx_kernel(const T* __restrict__ x,
                                  const T* __restrict__ dy,
                                  const T* __restrict__ w,
                                  const float* __restrict__ rstd,
                                  const T* __restrict__ plus,
                                  T* __restrict__ dx, long long R, int C) {
  __shared__ float scratch[kBlock / kWave];
  for (long long r = blockIdx.x; r < R; r += gridDim.x) {
    const T* xrow = x + r * C;
    const T* dyrow = dy + r * C;
    T* dxrow = dx + r * C;
    const float rs = rstd[r];
    constexpr int V = VecTraits<T>::kElems;
    const bool vec = (C % V) == 0;      // wave-uniform
    float s = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fx[V], fd[V], fw[V];
        vload16(xrow + cv * V, fx);
        vload16(dyrow + cv * V, fd);
        vload16(w + cv * V, fw);
#pragma unroll
        for (int q = 0; q < V; ++q) s = fmaf(fd[q] * fw[q], fx[q] * rs, s);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float dyw = (float)dyrow[c] * (float)w[c];
        float xhat = (float)xrow[c] * rs;
        s = fmaf(dyw, xhat, s);
      }
    }
    float m = block_reduce_sum(s, scratch) / C;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fx[V], fd[V], fw[V];
        vload16(xrow + cv * V, fx);
        vload16(dyrow + cv * V, fd);
        vload16(w + cv * V, fw);
#pragma unroll
        for (int q = 0; q < V; ++q)
          fd[q] = rs * (fd[q] * fw[q] - fx[q] * rs * m);
        if (PLUS) {
          float fp[V];
          vload16(plus + r * C + cv * V, fp);
#pragma unroll
          for (int q = 0; q < V; ++q) fd[q] += fp[q];
        }
        vstore16(dxrow + cv * V, fd);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float dyw = (float)dyrow[c] * (float)w[c];
        float xhat = (float)xrow[c] * rs;
        float v = rs * (dyw - xhat * m);
        if (PLUS) v += (float)plus[r * C + c];
        dxrow[c] = (T)v;
      }
    }
  }
}

// dw[c] = sum_r dy * xhat — column reduction parallel over channels AND
// rows: 64 channels x 4 row lanes per block, rows split over grid.y,
// LDS combine + atomicAdd (dw zero-initialized by the caller).
template <typename T>
__global__ void rms_bwd_dw_kernel(const T* __restrict__ x,
                                  const T* __restrict__ dy,
                                  const float* __restrict__ rstd,
                                  float* __restrict__ dw, long long R, int C,
                                  int rows_per_block) {
  __shared__ float sacc[4][64];
  const int c_local = threadIdx.x & 63;
  const int row_lane = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + c_local;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, R);
  float acc = 0.f;
  if (c < C) {
    for (long long r = r0 + row_lane; r < r1; r += 4)
      acc = fmaf((float)dy[r * C + c], (float)x[r * C + c] * rstd[r], acc);
  }
  sacc[row_lane][c_local] = acc;
  __syncthreads();
  if (row_lane == 0 && c < C) {
    float t = sacc[0][c_local] + sacc[1][c_local] + sacc[2][c_local] + sacc[3][c_local];
    if (gridDim.y == 1) dw[c] = t;
    else atomicAdd(&dw[c], t);
  }
}

// ---- RoPE (neox half-rotation) ---------------------------------------------
// x: [..., S, H, D] contiguous; pair (d, d+D/2) rotated by angle tables
// cos/sin [S, D/2] fp32. FWD=false applies the inverse rotation (backward).

template <typename T, bool INV>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, long long total,
                            int S, int H, int D) {
  const int half = D / 2;
  const long long pairs_per_s = (long long)H * half;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    // i indexes (b, s, h, d<half)
    const int d = (int)(i % half);
    const long long t = i / half;
    const int h = (int)(t % H);
    const long long t2 = t / H;
    const int s = (int)(t2 % S);
    const long long b = t2 / S;
    const long long base = ((b * S + s) * H + h) * (long long)D + d;
    const float c = cos_t[s * half + d];
    const float sn = sin_t[s * half + d];
    const float x1 = (float)x[base];
    const float x2 = (float)x[base + half];
    if (INV) {
      y[base] = (T)fmaf(x1, c, x2 * sn);
      y[base + half] = (T)fmaf(x2, c, -x1 * sn);
    } else {
      y[base] = (T)fmaf(x1, c, -x2 * sn);
      y[base + half] = (T)fmaf(x1, sn, x2 * c);
    }
  }
}

// ---- SwiGLU ----------------------------------------------------------------
// fwd: y = silu(a) * b;  bwd: da = dy*b*silu'(a), db = dy*silu(a)

template <typename T>
__global__ void silu_mul_fwd_kernel(const T* __restrict__ a,
                                    const T* __restrict__ b, T* __restrict__ y,
                                    long long n) {
  constexpr int V = VecTraits<T>::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    float fa[V], fb[V];
    vload16(a + i * V, fa);
    vload16(b + i * V, fb);
#pragma unroll
    for (int q = 0; q < V; ++q) {
      float sig = 1.f / (1.f + __expf(-fa[q]));
      fa[q] = fa[q] * sig * fb[q];
    }
    vstore16(y + i * V, fa);
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    float av = (float)a[i];
    float sig = 1.f / (1.f + __expf(-av));
    y[i] = (T)(av * sig * (float)b[i]);
  }
}

template <typename T>
__global__ void silu_mul_bwd_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ a,
                                    const T* __restrict__ b, T* __restrict__ da,
                                    T* __restrict__ db, long long n) {
  constexpr int V = VecTraits<T>::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    float fa[V], fb[V], fd[V], fda[V], fdb[V];
    vload16(a + i * V, fa);
    vload16(b + i * V, fb);
    vload16(dy + i * V, fd);
#pragma unroll
    for (int q = 0; q < V; ++q) {
      float sig = 1.f / (1.f + __expf(-fa[q]));
      float silu = fa[q] * sig;
      float dsilu = sig * fmaf(fa[q], 1.f - sig, 1.f);
      fda[q] = fd[q] * fb[q] * dsilu;
      fdb[q] = fd[q] * silu;
    }
    vstore16(da + i * V, fda);
    vstore16(db + i * V, fdb);
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    float av = (float)a[i];
    float dyv = (float)dy[i];
    float sig = 1.f / (1.f + __expf(-av));
    float silu = av * sig;
    float dsilu = sig * fmaf(av, 1.f - sig, 1.f);
    da[i] = (T)(dyv * (float)b[i] * dsilu);
    db[i] = (T)(dyv * silu);
  }
}

#define INST_LLAMA(T)                                                          \
  template __global__ void rms_fwd_kernel<T, false>(                           \
      const T*, const T*, const T*, T*, T*, float*, long long, int, float);    \
  template __global__ void rms_fwd_kernel<T, true>(                            \
      const T*, const T*, const T*, T*, T*, float*, long long, int, float);    \
  template __global__ void rms_bwd_dx_kernel<T, false>(                        \
      const T*, const T*, const T*, const float*, const T*, T*, long long,     \
      int);                                                                    \
  template __global__ void rms_bwd_dx_kernel<T, true>(                         \
      const T*, const T*, const T*, const float*, const T*, T*, long long,     \
      int);                                                                    \
  template __global__ void rms_bwd_dw_kernel<T>(const T*, const T*,            \
                                                const float*, float*,          \
                                                long long, int, int);          \
  template __global__ void rope_kernel<T, false>(const T*, T*, const float*,   \
                                                 const float*, long long, int, \
                                                 int, int);                    \
  template __global__ void rope_kernel<T, true>(const T*, T*, const float*,    \
                                                const float*, long long, int,  \
                                                int, int);                     \
  template __global__ void silu_mul_fwd_kernel<T>(const T*, const T*, T*,      \
                                                  long long);                  \
  template __global__ void silu_mul_bwd_kernel<T>(const T*, const T*,          \
                                                  const T*, T*, T*, long long);

INST_LLAMA(float)
INST_LLAMA(bf16)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_rms_fwd(bool is_bf16, const void* x, const void* res,
                    const void* w, void* y, void* zout, float* rstd,
                    long long R, int C, float eps, hipStream_t s) {
  const int grid = R < kMaxGrid ? (int)R : kMaxGrid;
  #define RMSF_CALL(T, A)                                                     \
    hipLaunchKernelGGL((rms_fwd_kernel<T, A>), dim3(grid), dim3(kBlock), 0,   \
                       s, (const T*)x, (const T*)res, (const T*)w, (T*)y,     \
                       (T*)zout, rstd, R, C, eps)
  if (is_bf16) { if (res) RMSF_CALL(bf16, true); else RMSF_CALL(bf16, false); }
  else { if (res) RMSF_CALL(float, true); else RMSF_CALL(float, false); }
  #undef RMSF_CALL
}

void launch_rms_bwd(bool is_bf16, const void* x, const void* dy, const void* w,
                    const float* rstd, const void* plus, void* dx, float* dw,
                    long long R, int C, hipStream_t s) {
  const int grid = R < kMaxGrid ? (int)R : kMaxGrid;
  const int cgrid = (C + 63) / 64;
  int target = 1024 / (cgrid > 0 ? cgrid : 1);
  if (target < 1) target = 1;
  long long rpb = (R + target - 1) / target;
  if (rpb < 128) rpb = 128;
  int gy = (int)((R + rpb - 1) / rpb);
  dim3 cg(cgrid, gy);
  #define RMSB_CALL(T, P)                                                     \
    hipLaunchKernelGGL((rms_bwd_dx_kernel<T, P>), dim3(grid), dim3(kBlock),   \
                       0, s, (const T*)x, (const T*)dy, (const T*)w, rstd,    \
                       (const T*)plus, (T*)dx, R, C)
  if (is_bf16) {
    if (plus) RMSB_CALL(bf16, true); else RMSB_CALL(bf16, false);
    hipLaunchKernelGGL(rms_bwd_dw_kernel<bf16>, cg, dim3(kBlock), 0,
                       s, (const bf16*)x, (const bf16*)dy, rstd, dw, R, C,
                       (int)rpb);
  } else {
    if (plus) RMSB_CALL(float, true); else RMSB_CALL(float, false);
    hipLaunchKernelGGL(rms_bwd_dw_kernel<float>, cg, dim3(kBlock), 0,
                       s, (const float*)x, (const float*)dy, rstd, dw, R, C,
                       (int)rpb);
  }
  #undef RMSB_CALL
}

void launch_rope(bool is_bf16, bool inverse, const void* x, void* y,
                 const float* cos_t, const float* sin_t, long long total_pairs,
                 int S, int H, int D, hipStream_t s) {
  const int grid = elementwise_grid(total_pairs);
  #define ROPE_CALL(T, I)                                                     \
    hipLaunchKernelGGL((rope_kernel<T, I>), dim3(grid), dim3(kBlock), 0, s,   \
                       (const T*)x, (T*)y, cos_t, sin_t, total_pairs, S, H, D)
  if (is_bf16) { if (inverse) ROPE_CALL(bf16, true); else ROPE_CALL(bf16, false); }
  else { if (inverse) ROPE_CALL(float, true); else ROPE_CALL(float, false); }
  #undef ROPE_CALL
}

void launch_silu_mul_fwd(bool is_bf16, const void* a, const void* b, void* y,
                         long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 4 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(silu_mul_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0,
                       s, (const bf16*)a, (const bf16*)b, (bf16*)y, n);
  else
    hipLaunchKernelGGL(silu_mul_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       s, (const float*)a, (const float*)b, (float*)y, n);
}

void launch_silu_mul_bwd(bool is_bf16, const void* dy, const void* a,
                         const void* b, void* da, void* db, long long n,
                         hipStream_t s) {
  const int grid = elementwise_grid(n / 4 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(silu_mul_bwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0,
                       s, (const bf16*)dy, (const bf16*)a, (const bf16*)b,
                       (bf16*)da, (bf16*)db, n);
  else
    hipLaunchKernelGGL(silu_mul_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       s, (const float*)dy, (const float*)a, (const float*)b,
                       (float*)da, (float*)db, n);
}
