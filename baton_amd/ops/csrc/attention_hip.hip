#include "hip/hip_runtime.h"
// Attention softmax kernels (gfx950).
//
// The transformer configs (BERT-base MLM, Llama-LoRA — BASELINE.json 3/4)
// compute attention as batched MFMA GEMMs (gemm.hip, grid.z = B*H) around
// this row softmax: y = softmax(scale * x [+ causal mask]) over the last
// dim. One block per row, online two-pass (max, then exp-sum), fused scale
// and causal masking — the score matrix is read once and written once.
// Backward fuses the dsoftmax reduction: dx = scale * y * (dy - sum(dy*y)).
#include "common.h"

template <typename T, bool CAUSAL>
__global__ void softmax_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   long long R, int C, float scale,
                                   int causal_seq) {
  __shared__ float scratch[kBlock / kWave];
  __shared__ float s_max;
  for (long long r = blockIdx.x; r < R; r += gridDim.x) {
    const T* row = x + r * C;
    T* yrow = y + r * C;
    const int qpos = CAUSAL ? (int)(r % causal_seq) : C - 1;
    constexpr int V = VecTraits<T>::kElems;
    // vector path needs enough vectors to keep a wave busy: at C=128 it
    // left 16 of 256 threads active and measured SLOWER than scalar
    const bool vec = (C % V) == 0 && C >= V * 64;
    float m = -INFINITY;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          if (cv * V + q <= qpos) m = fmaxf(m, f[q] * scale);
      }
    } else {
      for (int c = threadIdx.x; c <= qpos; c += blockDim.x)
        m = fmaxf(m, (float)row[c] * scale);
    }
    {
      const int lane = threadIdx.x & (kWave - 1);
      const int wid = threadIdx.x / kWave;
      float wm = wave_reduce_max(m);
      if (lane == 0) scratch[wid] = wm;
      __syncthreads();
      if (threadIdx.x == 0) {
        float t = -INFINITY;
        for (int i = 0; i < (int)(blockDim.x / kWave); ++i)
          t = fmaxf(t, scratch[i]);
        s_max = t;
      }
      __syncthreads();
    }
    const float m_all = s_max;
    float acc = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          if (cv * V + q <= qpos) acc += __expf(f[q] * scale - m_all);
      }
    } else {
      for (int c = threadIdx.x; c <= qpos; c += blockDim.x)
        acc += __expf((float)row[c] * scale - m_all);
    }
    float denom = block_reduce_sum(acc, scratch);
    const float inv = 1.f / denom;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          f[q] = cv * V + q <= qpos ? __expf(f[q] * scale - m_all) * inv : 0.f;
        vstore16(yrow + cv * V, f);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x)
        yrow[c] = (T)(c <= qpos ? __expf((float)row[c] * scale - m_all) * inv
                                : 0.f);
    }
    __syncthreads();
  }
}

// dx = scale * y * (dy - sum_c dy*y). Masked (y==0) columns get dx = 0.
template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ y,
                                   const T* __restrict__ dy, T* __restrict__ dx,
                                   long long R, int C, float scale) {
  __shared__ float scratch[kBlock / kWave];
  for (long long r = blockIdx.x; r < R; r += gridDim.x) {
    const T* yrow = y + r * C;
    const T* dyrow = dy + r * C;
    T* dxrow = dx + r * C;
    constexpr int V = VecTraits<T>::kElems;
    const bool vec = (C % V) == 0 && C >= V * 64;   // see fwd comment
    float acc = 0.f;
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fy[V], fd[V];
        vload16(yrow + cv * V, fy);
        vload16(dyrow + cv * V, fd);
#pragma unroll
        for (int q = 0; q < V; ++q) acc = fmaf(fd[q], fy[q], acc);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x)
        acc = fmaf((float)dyrow[c], (float)yrow[c], acc);
    }
    float dot = block_reduce_sum(acc, scratch);
    if (vec) {
      for (int cv = threadIdx.x; cv < C / V; cv += blockDim.x) {
        float fy[V], fd[V];
        vload16(yrow + cv * V, fy);
        vload16(dyrow + cv * V, fd);
#pragma unroll
        for (int q = 0; q < V; ++q) fd[q] = scale * fy[q] * (fd[q] - dot);
        vstore16(dxrow + cv * V, fd);
      }
    } else {
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float yv = (float)yrow[c];
        dxrow[c] = (T)(scale * yv * ((float)dyrow[c] - dot));
      }
    }
    __syncthreads();
  }
}

// Wave-per-row softmax for short rows (C <= 1024): four independent
// waves per block, wave-shuffle reductions, no block barriers — the
// row-per-block form leaves most of a 256-thread block idle at S=128.
template <typename T, bool CAUSAL>
__global__ __launch_bounds__(kBlock) void softmax_fwd_wave_kernel(
    const T* __restrict__ x, T* __restrict__ y, long long R, int C,
    float scale, int causal_seq) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  constexpr int V = VecTraits<T>::kElems;
  const bool vec = (C % V) == 0 && C >= V * 64;
  for (long long r = (long long)blockIdx.x * 4 + wid; r < R;
       r += (long long)gridDim.x * 4) {
    const T* row = x + r * C;
    T* yrow = y + r * C;
    const int qpos = CAUSAL ? (int)(r % causal_seq) : C - 1;
    float m = -INFINITY;
    if (vec) {
      for (int cv = lane; cv < C / V; cv += 64) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          if (cv * V + q <= qpos) m = fmaxf(m, f[q] * scale);
      }
    } else {
      for (int c = lane; c <= qpos; c += 64) m = fmaxf(m, (float)row[c] * scale);
    }
    m = wave_reduce_max(m);
    m = __shfl(m, 0, kWave);
    float acc = 0.f;
    if (vec) {
      for (int cv = lane; cv < C / V; cv += 64) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          if (cv * V + q <= qpos) acc += __expf(f[q] * scale - m);
      }
    } else {
      for (int c = lane; c <= qpos; c += 64) acc += __expf((float)row[c] * scale - m);
    }
    acc = wave_reduce_sum(acc);
    const float inv = 1.f / __shfl(acc, 0, kWave);
    if (vec) {
      for (int cv = lane; cv < C / V; cv += 64) {
        float f[V];
        vload16(row + cv * V, f);
#pragma unroll
        for (int q = 0; q < V; ++q)
          f[q] = cv * V + q <= qpos ? __expf(f[q] * scale - m) * inv : 0.f;
        vstore16(yrow + cv * V, f);
      }
    } else {
      for (int c = lane; c < C; c += 64)
        yrow[c] = (T)(c <= qpos ? __expf((float)row[c] * scale - m) * inv : 0.f);
    }
  }
}

template <typename T>
__global__ __launch_bounds__(kBlock) void softmax_bwd_wave_kernel(
    const T* __restrict__ y, const T* __restrict__ dy, T* __restrict__ dx,
    long long R, int C, float scale) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  constexpr int V = VecTraits<T>::kElems;
  const bool vec = (C % V) == 0 && C >= V * 64;
  for (long long r = (long long)blockIdx.x * 4 + wid; r < R;
       r += (long long)gridDim.x * 4) {
    const T* yrow = y + r * C;
    const T* dyrow = dy + r * C;
    T* dxrow = dx + r * C;
    float acc = 0.f;
    if (vec) {
      for (int cv = lane; cv < C / V; cv += 64) {
        float fy[V], fd[V];
        vload16(yrow + cv * V, fy);
        vload16(dyrow + cv * V, fd);
#pragma unroll
        for (int q = 0; q < V; ++q) acc = fmaf(fd[q], fy[q], acc);
      }
    } else {
      for (int c = lane; c < C; c += 64)
        acc = fmaf((float)dyrow[c], (float)yrow[c], acc);
    }
    acc = wave_reduce_sum(acc);
    const float dot = __shfl(acc, 0, kWave);
    if (vec) {
      for (int cv = lane; cv < C / V; cv += 64) {
        float fy[V], fd[V];
        vload16(yrow + cv * V, fy);
        vload16(dyrow + cv * V, fd);
#pragma unroll
        for (int q = 0; q < V; ++q) fd[q] = scale * fy[q] * (fd[q] - dot);
        vstore16(dxrow + cv * V, fd);
      }
    } else {
      for (int c = lane; c < C; c += 64) {
        float yv = (float)yrow[c];
        dxrow[c] = (T)(scale * yv * ((float)dyrow[c] - dot));
      }
    }
  }
}

#define INST_SM_WAVE(T)                                                      \
  template __global__ void softmax_fwd_wave_kernel<T, true>(                 \
      const T*, T*, long long, int, float, int);                             \
  template __global__ void softmax_fwd_wave_kernel<T, false>(                \
      const T*, T*, long long, int, float, int);                             \
  template __global__ void softmax_bwd_wave_kernel<T>(const T*, const T*,    \
                                                      T*, long long, int,    \
                                                      float);

INST_SM_WAVE(float)
INST_SM_WAVE(bf16)

#define INST_SM(T)                                                           \
  template __global__ void softmax_fwd_kernel<T, true>(const T*, T*,         \
                                                       long long, int, float, \
                                                       int);                 \
  template __global__ void softmax_fwd_kernel<T, false>(const T*, T*,        \
                                                        long long, int,      \
                                                        float, int);         \
  template __global__ void softmax_bwd_kernel<T>(const T*, const T*, T*,     \
                                                 long long, int, float);

INST_SM(float)
INST_SM(bf16)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_softmax_fwd(bool is_bf16, const void* x, void* y, long long R,
                        int C, float scale, int causal_seq, hipStream_t s) {
  if (C <= 1024) {   // wave-per-row: 4 rows/block, no block barriers
    const long long blocks = (R + 3) / 4;
    const int grid = blocks < kMaxGrid ? (int)blocks : kMaxGrid;
    #define SMW_CALL(T, CZ)                                                   \
      hipLaunchKernelGGL((softmax_fwd_wave_kernel<T, CZ>), dim3(grid),        \
                         dim3(kBlock), 0, s, (const T*)x, (T*)y, R, C, scale, \
                         causal_seq)
    if (is_bf16) {
      if (causal_seq > 0) SMW_CALL(bf16, true); else SMW_CALL(bf16, false);
    } else {
      if (causal_seq > 0) SMW_CALL(float, true); else SMW_CALL(float, false);
    }
    #undef SMW_CALL
    return;
  }
  const int grid = R < kMaxGrid ? (int)R : kMaxGrid;
  #define SM_CALL(T, CZ)                                                     \
    hipLaunchKernelGGL((softmax_fwd_kernel<T, CZ>), dim3(grid), dim3(kBlock), \
                       0, s, (const T*)x, (T*)y, R, C, scale, causal_seq)
  if (is_bf16) {
    if (causal_seq > 0) SM_CALL(bf16, true); else SM_CALL(bf16, false);
  } else {
    if (causal_seq > 0) SM_CALL(float, true); else SM_CALL(float, false);
  }
  #undef SM_CALL
}

void launch_softmax_bwd(bool is_bf16, const void* y, const void* dy, void* dx,
                        long long R, int C, float scale, hipStream_t s) {
  if (C <= 1024) {
    const long long blocks = (R + 3) / 4;
    const int grid = blocks < kMaxGrid ? (int)blocks : kMaxGrid;
    if (is_bf16)
      hipLaunchKernelGGL(softmax_bwd_wave_kernel<bf16>, dim3(grid),
                         dim3(kBlock), 0, s, (const bf16*)y, (const bf16*)dy,
                         (bf16*)dx, R, C, scale);
    else
      hipLaunchKernelGGL(softmax_bwd_wave_kernel<float>, dim3(grid),
                         dim3(kBlock), 0, s, (const float*)y, (const float*)dy,
                         (float*)dx, R, C, scale);
    return;
  }
  const int grid = R < kMaxGrid ? (int)R : kMaxGrid;
  if (is_bf16)
    hipLaunchKernelGGL(softmax_bwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0,
                       s, (const bf16*)y, (const bf16*)dy, (bf16*)dx, R, C,
                       scale);
  else
    hipLaunchKernelGGL(softmax_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       s, (const float*)y, (const float*)dy, (float*)dx, R, C,
                       scale);
}
