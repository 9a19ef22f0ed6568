// Normalization kernels (gfx950): LayerNorm + BatchNorm (NHWC), fwd + bwd.
//
// Not present in the reference (SURVEY.md §2.2 — its only model is
// Linear(10,1)) but required by the ResNet/BERT federated configs.
//
// LayerNorm: one block per row, block-reduced mean/var, saved (mean, rstd)
// for backward; dgamma/dbeta by a column-chunk reduction kernel (no
// atomics — each block owns a channel chunk and walks rows, coalesced in
// NHWC/row-major since consecutive lanes read consecutive channels).
// BatchNorm (training): per-channel stats by column-chunk reduction with a
// rows-split atomicAdd pass, then a vectorized normalize pass; eval mode
// normalizes with running stats.
#include "common.h"

// ---- LayerNorm forward -----------------------------------------------------

// ADD fuses the transformer residual add: z = x + res is written once and
// normalized in place (the at::add glue kernels).
template <typename T, bool ADD>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const T* __restrict__ res, const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              T* __restrict__ zout,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, int R, int C,
                              float eps) {
  __shared__ float scratch[kBlock / kWave];
  for (int r = blockIdx.x; r < R; r += gridDim.x) {
    const T* row = x + (long long)r * C;
    T* yrow = y + (long long)r * C;
    constexpr int V = VecTraits<T>::kElems;
    const bool vec = (C % V) == 0;      // wave-uniform
    float s = 0.f, ss = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V];
        vload16(row + cv * V, f);
        if (ADD) {
          float fr[V];
          vload16(res + (long long)r * C + cv * V, fr);
#pragma unroll
          for (int q = 0; q < V; ++q) f[q] += fr[q];
          vstore16(zout + (long long)r * C + cv * V, f);
        }
#pragma unroll
        for (int q = 0; q < V; ++q) {
          s += f[q];
          ss = fmaf(f[q], f[q], ss);
        }
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float v = (float)row[c];
        if (ADD) {
          v += (float)res[(long long)r * C + c];
          zout[(long long)r * C + c] = (T)v;
        }
        s += v;
        ss = fmaf(v, v, ss);
      }
    }
    float sum = block_reduce_sum(s, scratch);
    float sumsq = block_reduce_sum(ss, scratch);
    float mu = sum / C;
    float var = sumsq / C - mu * mu;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[r] = mu;
      rstd_out[r] = rstd;
    }
    const T* zrow = ADD ? zout + (long long)r * C : row;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V], fw[V], fb[V];
        vload16(zrow + cv * V, f);
        vload16(w + cv * V, fw);
        vload16(b + cv * V, fb);
#pragma unroll
        for (int q = 0; q < V; ++q)
          f[q] = fmaf((f[q] - mu) * rstd, fw[q], fb[q]);
        vstore16(yrow + cv * V, f);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float xhat = ((float)zrow[c] - mu) * rstd;
        yrow[c] = (T)fmaf(xhat, (float)w[c], (float)b[c]);
      }
    }
  }
}

// ---- LayerNorm backward (dx) ----------------------------------------------
// dx = rstd * (dyg - mean(dyg) - xhat * mean(dyg*xhat)), dyg = dy*gamma

// PLUS fuses the residual-stream gradient add (backward of z = x + res).
template <typename T, bool PLUS>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                 const T* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const T* __restrict__ plus,
                                 T* __restrict__ dx, int R, int C) {
  __shared__ float scratch[kBlock / kWave];
  for (int r = blockIdx.x; r < R; r += gridDim.x) {
    const T* xrow = x + (long long)r * C;
    const T* dyrow = dy + (long long)r * C;
    T* dxrow = dx + (long long)r * C;
    const float mu = mean[r], rs = rstd[r];
    constexpr int V = VecTraits<T>::kElems;
    const bool vec = (C % V) == 0;      // wave-uniform
    float s1 = 0.f, s2 = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fx[V], fd[V], fw[V];
        vload16(xrow + cv * V, fx);
        vload16(dyrow + cv * V, fd);
        vload16(w + cv * V, fw);
#pragma unroll
        for (int q = 0; q < V; ++q) {
          float dyg = fd[q] * fw[q];
          s1 += dyg;
          s2 = fmaf(dyg, (fx[q] - mu) * rs, s2);
        }
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float dyg = (float)dyrow[c] * (float)w[c];
        float xhat = ((float)xrow[c] - mu) * rs;
        s1 += dyg;
        s2 = fmaf(dyg, xhat, s2);
      }
    }
    float m1 = block_reduce_sum(s1, scratch) / C;
    float m2 = block_reduce_sum(s2, scratch) / C;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fx[V], fd[V], fw[V];
        vload16(xrow + cv * V, fx);
        vload16(dyrow + cv * V, fd);
        vload16(w + cv * V, fw);
#pragma unroll
        for (int q = 0; q < V; ++q) {
          float dyg = fd[q] * fw[q];
          float xhat = (fx[q] - mu) * rs;
          fd[q] = rs * (dyg - m1 - xhat * m2);
        }
        if (PLUS) {
          float fp[V];
          vload16(plus + (long long)r * C + cv * V, fp);
#pragma unroll
          for (int q = 0; q < V; ++q) fd[q] += fp[q];
        }
        vstore16(dxrow + cv * V, fd);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float dyg = (float)dyrow[c] * (float)w[c];
        float xhat = ((float)xrow[c] - mu) * rs;
        float v = rs * (dyg - m1 - xhat * m2);
        if (PLUS) v += (float)plus[(long long)r * C + c];
        dxrow[c] = (T)v;
      }
    }
  }
}

// dgamma[c] = sum_r dy*xhat ; dbeta[c] = sum_r dy. Column reduction with
// BOTH channel and row parallelism: a block covers 64 channels x 4 row
// lanes (all 256 threads busy even at C=64), rows additionally split over
// grid.y; 4-lane partials combine in LDS, then one atomicAdd per channel
// chunk (dw/db zero-initialized by the caller). The previous
// one-thread-per-channel form left ~3 blocks running on 256 CUs and was
// 49% of the BERT step.
template <typename T>
__global__ void ln_bwd_dwdb_kernel(const T* __restrict__ x,
                                   const T* __restrict__ dy,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   float* __restrict__ dw,
                                   float* __restrict__ db, int R, int C,
                                   int rows_per_block) {
  __shared__ float sw[4][64], sb[4][64];
  const int c_local = threadIdx.x & 63;
  const int row_lane = threadIdx.x >> 6;         // 0..3
  const int c = blockIdx.x * 64 + c_local;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, (long long)R);
  float acc_w = 0.f, acc_b = 0.f;
  if (c < C) {
    for (long long r = r0 + row_lane; r < r1; r += 4) {
      float d = (float)dy[r * C + c];
      float xhat = ((float)x[r * C + c] - mean[r]) * rstd[r];
      acc_w = fmaf(d, xhat, acc_w);
      acc_b += d;
    }
  }
  sw[row_lane][c_local] = acc_w;
  sb[row_lane][c_local] = acc_b;
  __syncthreads();
  if (row_lane == 0 && c < C) {
    float tw = sw[0][c_local] + sw[1][c_local] + sw[2][c_local] + sw[3][c_local];
    float tb = sb[0][c_local] + sb[1][c_local] + sb[2][c_local] + sb[3][c_local];
    if (gridDim.y == 1) {
      dw[c] = tw;
      db[c] = tb;
    } else {
      atomicAdd(&dw[c], tw);
      atomicAdd(&db[c], tb);
    }
  }
}

// ---- BatchNorm (NHWC: input viewed as [M rows, C channels]) ----------------

// Pass 1: per-channel sum & sumsq. Blocks tile (channel chunk, row chunk);
// row chunks atomically accumulate into sum[c]/sumsq[c] (caller zeroes).
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long long M, int C,
                                int rows_per_block) {
  __shared__ float s1[4][64], s2[4][64];
  const int c_local = threadIdx.x & 63;
  const int row_lane = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + c_local;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, M);
  float s = 0.f, ss = 0.f;
  if (c < C) {
    for (long long r = r0 + row_lane; r < r1; r += 4) {
      float v = (float)x[r * C + c];
      s += v;
      ss = fmaf(v, v, ss);
    }
  }
  s1[row_lane][c_local] = s;
  s2[row_lane][c_local] = ss;
  __syncthreads();
  if (row_lane == 0 && c < C) {
    atomicAdd(&sum[c], s1[0][c_local] + s1[1][c_local] + s1[2][c_local] + s1[3][c_local]);
    atomicAdd(&sumsq[c], s2[0][c_local] + s2[1][c_local] + s2[2][c_local] + s2[3][c_local]);
  }
}

// Vectorized stats (C % 64 == 0, the ResNet case): each thread owns an
// 8-channel 16-B run (G13 — the scalar walk above issues 2-B loads) and 32
// row-lanes stream rows; partials meet in a padded LDS tile, 64 threads
// column-reduce, then atomics. ~1.5x the scalar form at bench shapes.
template <typename T>
__global__ void bn_stats_vec_kernel(const T* __restrict__ x,
                                    float* __restrict__ sum,
                                    float* __restrict__ sumsq, long long M,
                                    int C, int rows_per_block) {
  using VT = VecTraits<T>;
  __shared__ float l1[32][65], l2[32][65];   // +1 pad: column reduce reads
  const int c8 = threadIdx.x & 7;
  const int rl = threadIdx.x >> 3;
  const int ch = blockIdx.x * 64 + c8 * 8;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, M);
  float s[8] = {}, ss[8] = {};
  for (long long r = r0 + rl; r < r1; r += 32) {
    typename VT::VecT v =
        *reinterpret_cast<const typename VT::VecT*>(&x[r * C + ch]);
    float f[VT::kElems];
    VT::to_float(v, f);
#pragma unroll
    for (int q = 0; q < VT::kElems; ++q) {
      s[q] += f[q];
      ss[q] = fmaf(f[q], f[q], ss[q]);
    }
  }
  // f32 input: kElems == 4 — thread covers 8 channels in two vectors
  if constexpr (VT::kElems == 4) {
    typename VT::VecT v2;
    for (long long r = r0 + rl; r < r1; r += 32) {
      v2 = *reinterpret_cast<const typename VT::VecT*>(&x[r * C + ch + 4]);
      float f[4];
      VT::to_float(v2, f);
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        s[4 + q] += f[q];
        ss[4 + q] = fmaf(f[q], f[q], ss[4 + q]);
      }
    }
  }
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    l1[rl][c8 * 8 + q] = s[q];
    l2[rl][c8 * 8 + q] = ss[q];
  }
  __syncthreads();
  if (threadIdx.x < 64) {
    float t1 = 0.f, t2 = 0.f;
#pragma unroll 8
    for (int r = 0; r < 32; ++r) {
      t1 += l1[r][threadIdx.x];
      t2 += l2[r][threadIdx.x];
    }
    atomicAdd(&sum[blockIdx.x * 64 + threadIdx.x], t1);
    atomicAdd(&sumsq[blockIdx.x * 64 + threadIdx.x], t2);
  }
}

template <typename T, bool RELU>
__global__ void bn_bwd_stats_vec_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const T* __restrict__ y_post, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ sum_dy,
    float* __restrict__ sum_dyx, long long M, int C, int rows_per_block) {
  using VT = VecTraits<T>;
  __shared__ float l1[32][65], l2[32][65];
  const int c8 = threadIdx.x & 7;
  const int rl = threadIdx.x >> 3;
  const int ch = blockIdx.x * 64 + c8 * 8;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, M);
  float mu[8], rs[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    mu[q] = mean[ch + q];
    rs[q] = rstd[ch + q];
  }
  float s1v[8] = {}, s2v[8] = {};
  constexpr int V = VT::kElems;
  for (long long r = r0 + rl; r < r1; r += 32) {
#pragma unroll
    for (int part = 0; part < 8 / V; ++part) {
      const long long base = r * C + ch + part * V;
      float fx[V], fd[V], fy[V];
      typename VT::VecT vx =
          *reinterpret_cast<const typename VT::VecT*>(&x[base]);
      typename VT::VecT vd =
          *reinterpret_cast<const typename VT::VecT*>(&dy[base]);
      VT::to_float(vx, fx);
      VT::to_float(vd, fd);
      if (RELU) {
        typename VT::VecT vy =
            *reinterpret_cast<const typename VT::VecT*>(&y_post[base]);
        VT::to_float(vy, fy);
      }
#pragma unroll
      for (int q = 0; q < V; ++q) {
        float d = fd[q];
        if (RELU) d = fy[q] > 0.f ? d : 0.f;
        const int qq = part * V + q;
        float xhat = (fx[q] - mu[qq]) * rs[qq];
        s1v[qq] += d;
        s2v[qq] = fmaf(d, xhat, s2v[qq]);
      }
    }
  }
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    l1[rl][c8 * 8 + q] = s1v[q];
    l2[rl][c8 * 8 + q] = s2v[q];
  }
  __syncthreads();
  if (threadIdx.x < 64) {
    float t1 = 0.f, t2 = 0.f;
#pragma unroll 8
    for (int r = 0; r < 32; ++r) {
      t1 += l1[r][threadIdx.x];
      t2 += l2[r][threadIdx.x];
    }
    atomicAdd(&sum_dy[blockIdx.x * 64 + threadIdx.x], t1);
    atomicAdd(&sum_dyx[blockIdx.x * 64 + threadIdx.x], t2);
  }
}

// Pass 2 (one small block over C): finalize mean/rstd, update running stats.
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, long long M,
                                   int C, float eps, float momentum) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mu = sum[c] / (float)M;
  float var = sumsq[c] / (float)M - mu * mu;  // biased (torch normalizes with biased)
  var = fmaxf(var, 0.f);
  mean[c] = mu;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    float unbiased = M > 1 ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = fmaf(momentum, mu - running_mean[c], running_mean[c]);
    running_var[c] = fmaf(momentum, unbiased - running_var[c], running_var[c]);
  }
}

// Pass 3: y = (x - mean) * rstd * gamma + beta, optional fused ReLU;
// ADDIN fuses the ResNet residual join (y = relu(bn(x) + res)) — the
// standalone add_relu pass and one full activation read disappear.
template <typename T, bool RELU, bool ADDIN = false>
__global__ void bn_norm_kernel(const T* __restrict__ x,
                               const T* __restrict__ res,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta,
                               T* __restrict__ y, long long M, int C) {
  const long long total = M * C;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(i % C);
    float v = ((float)x[i] - mean[c]) * rstd[c];
    v = fmaf(v, gamma[c], beta[c]);
    if (ADDIN) v += (float)res[i];
    if (RELU) v = fmaxf(v, 0.f);
    y[i] = (T)v;
  }
}

// Vectorized normalize (C % kElems == 0): y = x*K1[c] + K2[c] — the two
// per-channel constants are precomputed into LDS once per block, then the
// main loop is pure 16-B traffic (the scalar form costs a %C and four
// scalar param loads per ELEMENT).
template <typename T, bool RELU, bool ADDIN = false>
__global__ void bn_norm_vec_kernel(const T* __restrict__ x,
                                   const T* __restrict__ res,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   T* __restrict__ y, long long M, int C) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  extern __shared__ float bn_sm[];
  float* K1 = bn_sm;          // rstd*gamma
  float* K2 = bn_sm + C;      // beta - mean*rstd*gamma
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float g = rstd[c] * gamma[c];
    K1[c] = g;
    K2[c] = fmaf(-mean[c], g, beta[c]);
  }
  __syncthreads();
  const long long nvec = M * C / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    const int c0 = (int)((i * V) % C);
    typename VT::VecT v = reinterpret_cast<const typename VT::VecT*>(x)[i];
    float f[V];
    VT::to_float(v, f);
    float fr[V];
    if (ADDIN) {
      typename VT::VecT vr = reinterpret_cast<const typename VT::VecT*>(res)[i];
      VT::to_float(vr, fr);
    }
#pragma unroll
    for (int q = 0; q < V; ++q) {
      f[q] = fmaf(f[q], K1[c0 + q], K2[c0 + q]);
      if (ADDIN) f[q] += fr[q];
      if (RELU) f[q] = fmaxf(f[q], 0.f);
    }
    typename VT::VecT o;
    VT::from_float(f, o);
    reinterpret_cast<typename VT::VecT*>(y)[i] = o;
  }
}

// Vectorized dx: dx = K1[c]*d - x*K3[c] + K4[c] (d optionally ReLU-masked
// by y_post), constants folded per channel in LDS.
// DRES additionally writes the residual-branch gradient of the fused
// y = relu(bn(x) + res) forward: dres = relu-masked dy (same mask as dx's).
template <typename T, bool RELU, bool DRES = false>
__global__ void bn_bwd_dx_vec_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const T* __restrict__ y_post, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ sum_dy, const float* __restrict__ sum_dyx,
    T* __restrict__ dx, T* __restrict__ dres, long long M, int C) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  extern __shared__ float bn_sm[];
  float* K1 = bn_sm;           // rstd*gamma
  float* K3 = bn_sm + C;       // rstd*gamma * rstd * sum_dyx/M
  float* K4 = bn_sm + 2 * C;   // mean*K3 - K1*sum_dy/M
  const float invM = 1.f / (float)M;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float g = rstd[c] * gamma[c];
    const float k3 = g * rstd[c] * sum_dyx[c] * invM;
    K1[c] = g;
    K3[c] = k3;
    K4[c] = fmaf(mean[c], k3, -g * sum_dy[c] * invM);
  }
  __syncthreads();
  const long long nvec = M * C / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    const int c0 = (int)((i * V) % C);
    float fx[V], fd[V], fy[V];
    typename VT::VecT vx = reinterpret_cast<const typename VT::VecT*>(x)[i];
    typename VT::VecT vd = reinterpret_cast<const typename VT::VecT*>(dy)[i];
    VT::to_float(vx, fx);
    VT::to_float(vd, fd);
    if (RELU) {
      typename VT::VecT vy =
          reinterpret_cast<const typename VT::VecT*>(y_post)[i];
      VT::to_float(vy, fy);
    }
    float fdr[V];
#pragma unroll
    for (int q = 0; q < V; ++q) {
      float d = fd[q];
      if (RELU) d = fy[q] > 0.f ? d : 0.f;
      if (DRES) fdr[q] = d;
      fd[q] = fmaf(d, K1[c0 + q], fmaf(-fx[q], K3[c0 + q], K4[c0 + q]));
    }
    typename VT::VecT o;
    VT::from_float(fd, o);
    reinterpret_cast<typename VT::VecT*>(dx)[i] = o;
    if (DRES) {
      typename VT::VecT orr;
      VT::from_float(fdr, orr);
      reinterpret_cast<typename VT::VecT*>(dres)[i] = orr;
    }
  }
}

// Backward pass 1: per-channel sum(dy) and sum(dy*xhat). If RELU_FUSED, dy
// is masked by (y > 0) first (y = the post-ReLU output, also masked in dx).
template <typename T, bool RELU>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y_post,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ sum_dy,
                                    float* __restrict__ sum_dyx, long long M,
                                    int C, int rows_per_block) {
  __shared__ float l1[4][64], l2[4][64];
  const int c_local = threadIdx.x & 63;
  const int row_lane = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + c_local;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, M);
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    const float mu = mean[c], rs = rstd[c];
    for (long long r = r0 + row_lane; r < r1; r += 4) {
      float d = (float)dy[r * C + c];
      if (RELU) d = ((float)y_post[r * C + c] > 0.f) ? d : 0.f;
      float xhat = ((float)x[r * C + c] - mu) * rs;
      s1 += d;
      s2 = fmaf(d, xhat, s2);
    }
  }
  l1[row_lane][c_local] = s1;
  l2[row_lane][c_local] = s2;
  __syncthreads();
  if (row_lane == 0 && c < C) {
    atomicAdd(&sum_dy[c], l1[0][c_local] + l1[1][c_local] + l1[2][c_local] + l1[3][c_local]);
    atomicAdd(&sum_dyx[c], l2[0][c_local] + l2[1][c_local] + l2[2][c_local] + l2[3][c_local]);
  }
}

// Backward pass 2: dx = rstd*gamma*(dy - sum_dy/M - xhat*sum_dyx/M)
template <typename T, bool RELU>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ x,
                                 const T* __restrict__ dy,
                                 const T* __restrict__ y_post,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ sum_dy,
                                 const float* __restrict__ sum_dyx,
                                 T* __restrict__ dx, long long M, int C) {
  const long long total = M * C;
  const float invM = 1.f / (float)M;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(i % C);
    float d = (float)dy[i];
    if (RELU) d = ((float)y_post[i] > 0.f) ? d : 0.f;
    float xhat = ((float)x[i] - mean[c]) * rstd[c];
    float g = rstd[c] * gamma[c];
    dx[i] = (T)(g * (d - sum_dy[c] * invM - xhat * sum_dyx[c] * invM));
  }
}

#define INST_BN(T)                                                              \
  template __global__ void bn_stats_kernel<T>(const T*, float*, float*,         \
                                              long long, int, int);             \
  template __global__ void bn_norm_kernel<T, true>(                             \
      const T*, const T*, const float*, const float*, const float*,             \
      const float*, T*, long long, int);                                        \
  template __global__ void bn_norm_kernel<T, false>(                            \
      const T*, const T*, const float*, const float*, const float*,             \
      const float*, T*, long long, int);                                        \
  template __global__ void bn_norm_kernel<T, true, true>(                       \
      const T*, const T*, const float*, const float*, const float*,             \
      const float*, T*, long long, int);                                        \
  template __global__ void bn_bwd_stats_kernel<T, true>(                        \
      const T*, const T*, const T*, const float*, const float*, float*, float*, \
      long long, int, int);                                                     \
  template __global__ void bn_bwd_stats_kernel<T, false>(                       \
      const T*, const T*, const T*, const float*, const float*, float*, float*, \
      long long, int, int);                                                     \
  template __global__ void bn_bwd_dx_kernel<T, true>(                           \
      const T*, const T*, const T*, const float*, const float*, const float*,   \
      const float*, const float*, T*, long long, int);                          \
  template __global__ void bn_bwd_dx_kernel<T, false>(                          \
      const T*, const T*, const T*, const float*, const float*, const float*,   \
      const float*, const float*, T*, long long, int);

INST_BN(float)
INST_BN(bf16)

#define INST_LN(T)                                                             \
  template __global__ void ln_fwd_kernel<T, false>(                            \
      const T*, const T*, const T*, const T*, T*, T*, float*, float*, int,     \
      int, float);                                                             \
  template __global__ void ln_fwd_kernel<T, true>(                             \
      const T*, const T*, const T*, const T*, T*, T*, float*, float*, int,     \
      int, float);                                                             \
  template __global__ void ln_bwd_dx_kernel<T, false>(                         \
      const T*, const T*, const T*, const float*, const float*, const T*, T*,  \
      int, int);                                                               \
  template __global__ void ln_bwd_dx_kernel<T, true>(                          \
      const T*, const T*, const T*, const float*, const float*, const T*, T*,  \
      int, int);                                                               \
  template __global__ void ln_bwd_dwdb_kernel<T>(const T*, const T*,           \
                                                 const float*, const float*,   \
                                                 float*, float*, int, int,     \
                                                 int);

INST_LN(float)
INST_LN(bf16)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

static int row_grid(long long rows) { return rows < kMaxGrid ? (int)rows : kMaxGrid; }

void launch_ln_fwd(bool is_bf16, const void* x, const void* res, const void* w,
                   const void* b, void* y, void* zout, float* mean,
                   float* rstd, int R, int C, float eps, hipStream_t s) {
  #define LNF_CALL(T, A)                                                      \
    hipLaunchKernelGGL((ln_fwd_kernel<T, A>), dim3(row_grid(R)),              \
                       dim3(kBlock), 0, s, (const T*)x, (const T*)res,        \
                       (const T*)w, (const T*)b, (T*)y, (T*)zout, mean, rstd, \
                       R, C, eps)
  if (is_bf16) { if (res) LNF_CALL(bf16, true); else LNF_CALL(bf16, false); }
  else { if (res) LNF_CALL(float, true); else LNF_CALL(float, false); }
  #undef LNF_CALL
}

void launch_ln_bwd_dx(bool is_bf16, const void* x, const void* dy, const void* w,
                      const float* mean, const float* rstd, const void* plus,
                      void* dx, int R, int C, hipStream_t s) {
  #define LNB_CALL(T, P)                                                      \
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, P>), dim3(row_grid(R)),           \
                       dim3(kBlock), 0, s, (const T*)x, (const T*)dy,         \
                       (const T*)w, mean, rstd, (const T*)plus, (T*)dx, R, C)
  if (is_bf16) { if (plus) LNB_CALL(bf16, true); else LNB_CALL(bf16, false); }
  else { if (plus) LNB_CALL(float, true); else LNB_CALL(float, false); }
  #undef LNB_CALL
}

void launch_ln_bwd_dwdb(bool is_bf16, const void* x, const void* dy,
                        const float* mean, const float* rstd, float* dw,
                        float* db, int R, int C, hipStream_t s) {
  const int cgrid = (C + 63) / 64;
  int target = 1024 / (cgrid > 0 ? cgrid : 1);
  if (target < 1) target = 1;
  int rpb = (R + target - 1) / target;
  if (rpb < 128) rpb = 128;
  int gy = (R + rpb - 1) / rpb;
  dim3 grid(cgrid, gy);
  if (is_bf16)
    hipLaunchKernelGGL(ln_bwd_dwdb_kernel<bf16>, grid, dim3(kBlock), 0, s,
                       (const bf16*)x, (const bf16*)dy, mean, rstd, dw, db, R,
                       C, rpb);
  else
    hipLaunchKernelGGL(ln_bwd_dwdb_kernel<float>, grid, dim3(kBlock), 0,
                       s, (const float*)x, (const float*)dy, mean, rstd, dw, db,
                       R, C, rpb);
}

// rows_per_block tuned so grid.y gives ~8 blocks/CU worth of parallelism
static void bn_rows_split(long long M, int C, int* rows_per_block, int* grid_y) {
  int col_blocks = (C + 63) / 64;
  long long target_blocks = 2048 / (col_blocks > 0 ? col_blocks : 1);
  if (target_blocks < 1) target_blocks = 1;
  long long rpb = (M + target_blocks - 1) / target_blocks;
  if (rpb < 64) rpb = 64;
  *rows_per_block = (int)rpb;
  *grid_y = (int)((M + rpb - 1) / rpb);
}

void launch_bn_stats(bool is_bf16, const void* x, float* sum, float* sumsq,
                     long long M, int C, hipStream_t s) {
  int rpb, gy;
  bn_rows_split(M, C, &rpb, &gy);
  dim3 grid((C + 63) / 64, gy);
  if ((C % 64) == 0) {   // vectorized 16-B path (every ResNet BN width)
    if (is_bf16)
      hipLaunchKernelGGL(bn_stats_vec_kernel<bf16>, grid, dim3(kBlock), 0, s,
                         (const bf16*)x, sum, sumsq, M, C, rpb);
    else
      hipLaunchKernelGGL(bn_stats_vec_kernel<float>, grid, dim3(kBlock), 0, s,
                         (const float*)x, sum, sumsq, M, C, rpb);
    return;
  }
  if (is_bf16)
    hipLaunchKernelGGL(bn_stats_kernel<bf16>, grid, dim3(kBlock), 0, s,
                       (const bf16*)x, sum, sumsq, M, C, rpb);
  else
    hipLaunchKernelGGL(bn_stats_kernel<float>, grid, dim3(kBlock), 0, s,
                       (const float*)x, sum, sumsq, M, C, rpb);
}

void launch_bn_finalize(const float* sum, const float* sumsq, float* mean,
                        float* rstd, float* running_mean, float* running_var,
                        long long M, int C, float eps, float momentum,
                        hipStream_t s) {
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + kBlock - 1) / kBlock),
                     dim3(kBlock), 0, s, sum, sumsq, mean, rstd, running_mean,
                     running_var, M, C, eps, momentum);
}

void launch_bn_norm(bool is_bf16, bool relu, const void* x, const void* res,
                    const float* mean, const float* rstd, const float* gamma,
                    const float* beta, void* y, long long M, int C,
                    hipStream_t s) {
  const int grid = elementwise_grid(M * C / 4 + 1);
  const int velems = is_bf16 ? 8 : 4;
  if ((C % velems) == 0 && C <= 6144) {   // vectorized + LDS constants
    const size_t smem = 2u * C * sizeof(float);
    #define BN_NV(T, R, A)                                                   \
      hipLaunchKernelGGL((bn_norm_vec_kernel<T, R, A>), dim3(grid),          \
                         dim3(kBlock), smem, s, (const T*)x, (const T*)res,  \
                         mean, rstd, gamma, beta, (T*)y, M, C)
    if (is_bf16) {
      if (res) BN_NV(bf16, true, true);
      else if (relu) BN_NV(bf16, true, false);
      else BN_NV(bf16, false, false);
    } else {
      if (res) BN_NV(float, true, true);
      else if (relu) BN_NV(float, true, false);
      else BN_NV(float, false, false);
    }
    #undef BN_NV
    return;
  }
  #define BN_NORM(T, R, A)                                                  \
    hipLaunchKernelGGL((bn_norm_kernel<T, R, A>), dim3(grid), dim3(kBlock), \
                       0, s, (const T*)x, (const T*)res, mean, rstd, gamma, \
                       beta, (T*)y, M, C)
  if (is_bf16) {
    if (res) BN_NORM(bf16, true, true);
    else if (relu) BN_NORM(bf16, true, false);
    else BN_NORM(bf16, false, false);
  } else {
    if (res) BN_NORM(float, true, true);
    else if (relu) BN_NORM(float, true, false);
    else BN_NORM(float, false, false);
  }
  #undef BN_NORM
}

void launch_bn_bwd_stats(bool is_bf16, bool relu, const void* x, const void* dy,
                         const void* y_post, const float* mean,
                         const float* rstd, float* sum_dy, float* sum_dyx,
                         long long M, int C, hipStream_t s) {
  int rpb, gy;
  bn_rows_split(M, C, &rpb, &gy);
  dim3 grid((C + 63) / 64, gy);
  if ((C % 64) == 0) {   // vectorized 16-B path
    #define BN_BSV(T, R)                                                       \
      hipLaunchKernelGGL((bn_bwd_stats_vec_kernel<T, R>), grid, dim3(kBlock), \
                         0, s, (const T*)x, (const T*)dy, (const T*)y_post,   \
                         mean, rstd, sum_dy, sum_dyx, M, C, rpb)
    if (is_bf16) { if (relu) BN_BSV(bf16, true); else BN_BSV(bf16, false); }
    else { if (relu) BN_BSV(float, true); else BN_BSV(float, false); }
    #undef BN_BSV
    return;
  }
  #define BN_BS(T, R)                                                        \
    hipLaunchKernelGGL((bn_bwd_stats_kernel<T, R>), grid, dim3(kBlock), 0, s, \
                       (const T*)x, (const T*)dy, (const T*)y_post, mean,     \
                       rstd, sum_dy, sum_dyx, M, C, rpb)
  if (is_bf16) { if (relu) BN_BS(bf16, true); else BN_BS(bf16, false); }
  else { if (relu) BN_BS(float, true); else BN_BS(float, false); }
  #undef BN_BS
}

void launch_bn_bwd_dx(bool is_bf16, bool relu, const void* x, const void* dy,
                      const void* y_post, const float* mean, const float* rstd,
                      const float* gamma, const float* sum_dy,
                      const float* sum_dyx, void* dx, void* dres, long long M,
                      int C, hipStream_t s) {
  const int grid = elementwise_grid(M * C / 4 + 1);
  const int velems = is_bf16 ? 8 : 4;
  if ((C % velems) == 0 && C <= 4096) {   // vectorized + LDS constants
    const size_t smem = 3u * C * sizeof(float);
    #define BN_DXV(T, R, D)                                                   \
      hipLaunchKernelGGL((bn_bwd_dx_vec_kernel<T, R, D>), dim3(grid),         \
                         dim3(kBlock), smem, s, (const T*)x, (const T*)dy,    \
                         (const T*)y_post, mean, rstd, gamma, sum_dy,         \
                         sum_dyx, (T*)dx, (T*)dres, M, C)
    if (is_bf16) {
      if (dres) BN_DXV(bf16, true, true);
      else if (relu) BN_DXV(bf16, true, false);
      else BN_DXV(bf16, false, false);
    } else {
      if (dres) BN_DXV(float, true, true);
      else if (relu) BN_DXV(float, true, false);
      else BN_DXV(float, false, false);
    }
    #undef BN_DXV
    return;
  }
  #define BN_DX(T, R)                                                       \
    hipLaunchKernelGGL((bn_bwd_dx_kernel<T, R>), dim3(grid), dim3(kBlock),  \
                       0, s, (const T*)x, (const T*)dy, (const T*)y_post,   \
                       mean, rstd, gamma, sum_dy, sum_dyx, (T*)dx, M, C)
  if (is_bf16) { if (relu) BN_DX(bf16, true); else BN_DX(bf16, false); }
  else { if (relu) BN_DX(float, true); else BN_DX(float, false); }
  #undef BN_DX
}
