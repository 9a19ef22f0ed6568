// FedAvg arithmetic kernels (gfx950): the pre-scale / scale / axpby ops of
// the RCCL data plane (SURVEY.md §2.2 row "FedAvg aggregation").
//
// theta-traffic pattern: each GPU-client writes (n_i/N) * theta_i into the
// fp32 reduce buffer (scale_cast), RCCL reduces to root, root broadcasts,
// every client installs the result back into its (possibly bf16) params
// (cast_copy). All streaming ops: 16 B/lane, grid-stride.
#include "common.h"

// dst(f32) = alpha * src(T)
template <typename T>
__global__ void scale_cast_kernel(float* __restrict__ dst,
                                  const T* __restrict__ src, long long n,
                                  float alpha) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT sv = reinterpret_cast<const typename VT::VecT*>(src)[i];
    float f[V];
    VT::to_float(sv, f);
#pragma unroll
    for (int q = 0; q < V / 4; ++q) {
      f32x4 o;
#pragma unroll
      for (int k = 0; k < 4; ++k) o[k] = alpha * f[q * 4 + k];
      reinterpret_cast<f32x4*>(dst)[i * (V / 4) + q] = o;
    }
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    dst[i] = alpha * (float)src[i];
}

// dst(T) = src(f32)
template <typename T>
__global__ void cast_copy_kernel(T* __restrict__ dst,
                                 const float* __restrict__ src, long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    float f[V];
#pragma unroll
    for (int q = 0; q < V / 4; ++q) {
      f32x4 s = reinterpret_cast<const f32x4*>(src)[i * (V / 4) + q];
#pragma unroll
      for (int k = 0; k < 4; ++k) f[q * 4 + k] = s[k];
    }
    typename VT::VecT o;
    VT::from_float(f, o);
    reinterpret_cast<typename VT::VecT*>(dst)[i] = o;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    dst[i] = (T)src[i];
}

// y = a*x + b*y (fp32) — the generalized aggregation update
__global__ void axpby_kernel(float* __restrict__ y, const float* __restrict__ x,
                             long long n, float a, float b) {
  const long long nvec = n / 4;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    f32x4 xv = reinterpret_cast<const f32x4*>(x)[i];
    f32x4 yv = reinterpret_cast<const f32x4*>(y)[i];
#pragma unroll
    for (int k = 0; k < 4; ++k) yv[k] = fmaf(a, xv[k], b * yv[k]);
    reinterpret_cast<f32x4*>(y)[i] = yv;
  }
  for (long long i = nvec * 4 + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    y[i] = fmaf(a, x[i], b * y[i]);
}

template __global__ void scale_cast_kernel<float>(float*, const float*,
                                                  long long, float);
template __global__ void scale_cast_kernel<bf16>(float*, const bf16*, long long,
                                                 float);
template __global__ void cast_copy_kernel<float>(float*, const float*, long long);
template __global__ void cast_copy_kernel<bf16>(bf16*, const float*, long long);

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_scale_cast(bool src_bf16, float* dst, const void* src, long long n,
                       float alpha, hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (src_bf16)
    hipLaunchKernelGGL(scale_cast_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       dst, (const bf16*)src, n, alpha);
  else
    hipLaunchKernelGGL(scale_cast_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       dst, (const float*)src, n, alpha);
}

void launch_cast_copy(bool dst_bf16, void* dst, const float* src, long long n,
                      hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (dst_bf16)
    hipLaunchKernelGGL(cast_copy_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (bf16*)dst, src, n);
  else
    hipLaunchKernelGGL(cast_copy_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (float*)dst, src, n);
}

void launch_axpby(float* y, const float* x, long long n, float a, float b,
                  hipStream_t s) {
  hipLaunchKernelGGL(axpby_kernel, dim3(elementwise_grid(n / 4 + 1)),
                     dim3(kBlock), 0, s, y, x, n, a, b);
}

// ---- bf16/f32 matrix transpose --------------------------------------------
// [R,C] -> [C,R], 64x64 LDS tiles (+8-element pad kills bank conflicts),
// 16-B vector loads AND stores. Used by the GEMM router: transposing an
// operand once (memory-bound, ~us) moves NN/TN GEMMs onto the glds NT path
// (645-807 TF vs 160-460 measured at training shapes).
template <typename T>
__global__ __launch_bounds__(256) void transpose_kernel(
    const T* __restrict__ in, T* __restrict__ out, int R, int C) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int TILE = 64;
  __shared__ T tile[TILE][TILE + ELEMS];
  const int r0 = blockIdx.y * TILE, c0 = blockIdx.x * TILE;
  using VT = typename VecTraits<T>::VecT;
  // load: each thread one 16-B run along C
  {
    constexpr int TPR = TILE / ELEMS;              // threads per row
    for (int idx = threadIdx.x; idx < TILE * TPR; idx += 256) {
      int r = idx / TPR, cc = (idx % TPR) * ELEMS;
      VT v;
      T* vp = reinterpret_cast<T*>(&v);
      if (r0 + r < R && c0 + cc + ELEMS <= C) {
        v = *reinterpret_cast<const VT*>(&in[(long long)(r0 + r) * C + c0 + cc]);
      } else {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j)
          vp[j] = (r0 + r < R && c0 + cc + j < C)
                      ? in[(long long)(r0 + r) * C + c0 + cc + j] : (T)0.f;
      }
      *reinterpret_cast<VT*>(&tile[r][cc]) = v;
    }
  }
  __syncthreads();
  // store: each thread one 16-B run along R (reads a strided LDS column)
  {
    constexpr int TPC = TILE / ELEMS;
    for (int idx = threadIdx.x; idx < TILE * TPC; idx += 256) {
      int c = idx / TPC, rr = (idx % TPC) * ELEMS;
      if (c0 + c >= C) continue;
      VT v;
      T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
      for (int j = 0; j < ELEMS; ++j) vp[j] = tile[rr + j][c];
      if (r0 + rr + ELEMS <= R) {
        *reinterpret_cast<VT*>(&out[(long long)(c0 + c) * R + r0 + rr]) = v;
      } else {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j)
          if (r0 + rr + j < R) out[(long long)(c0 + c) * R + r0 + rr + j] = vp[j];
      }
    }
  }
}

template __global__ void transpose_kernel<bf16>(const bf16*, bf16*, int, int);
template __global__ void transpose_kernel<float>(const float*, float*, int, int);

void launch_transpose(bool is_bf16, const void* in, void* out, int R, int C,
                      hipStream_t s) {
  dim3 grid((C + 63) / 64, (R + 63) / 64);
  if (is_bf16)
    hipLaunchKernelGGL(transpose_kernel<bf16>, grid, dim3(256), 0, s,
                       (const bf16*)in, (bf16*)out, R, C);
  else
    hipLaunchKernelGGL(transpose_kernel<float>, grid, dim3(256), 0, s,
                       (const float*)in, (float*)out, R, C);
}

// ---- column sum (Linear bias gradient) -------------------------------------
// db[c] = sum_r dy[r, c] — 64-channel x 4-row-lane blocks, rows split over
// grid.y, fp32 atomics (same shape discipline as the norm dgamma kernels).
template <typename T>
__global__ void colsum_kernel(const T* __restrict__ x, float* __restrict__ out,
                              long long R, int C, int rows_per_block) {
  __shared__ float sacc[4][64];
  const int c_local = threadIdx.x & 63;
  const int row_lane = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + c_local;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, R);
  float acc = 0.f;
  if (c < C) {
    for (long long r = r0 + row_lane; r < r1; r += 4)
      acc += (float)x[r * C + c];
  }
  sacc[row_lane][c_local] = acc;
  __syncthreads();
  if (row_lane == 0 && c < C) {
    float t = sacc[0][c_local] + sacc[1][c_local] + sacc[2][c_local] + sacc[3][c_local];
    if (gridDim.y == 1) out[c] = t;
    else atomicAdd(&out[c], t);
  }
}

// Vectorized colsum (C % 64 == 0): 8-channel 16-B runs per lane, 32 row
// lanes, padded-LDS column reduce (same discipline as bn_stats_vec).
template <typename T>
__global__ void colsum_vec_kernel(const T* __restrict__ x,
                                  float* __restrict__ out, long long R, int C,
                                  int rows_per_block) {
  using VT = VecTraits<T>;
  __shared__ float sacc[32][65];
  const int c8 = threadIdx.x & 7;
  const int rl = threadIdx.x >> 3;
  const int ch = blockIdx.x * 64 + c8 * 8;
  const long long r0 = (long long)blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, R);
  float acc[8] = {};
  constexpr int V = VT::kElems;
  for (long long r = r0 + rl; r < r1; r += 32) {
#pragma unroll
    for (int part = 0; part < 8 / V; ++part) {
      float f[V];
      typename VT::VecT v = *reinterpret_cast<const typename VT::VecT*>(
          &x[r * C + ch + part * V]);
      VT::to_float(v, f);
#pragma unroll
      for (int q = 0; q < V; ++q) acc[part * V + q] += f[q];
    }
  }
#pragma unroll
  for (int q = 0; q < 8; ++q) sacc[rl][c8 * 8 + q] = acc[q];
  __syncthreads();
  if (threadIdx.x < 64) {
    float t = 0.f;
#pragma unroll 8
    for (int r = 0; r < 32; ++r) t += sacc[r][threadIdx.x];
    if (gridDim.y == 1) out[blockIdx.x * 64 + threadIdx.x] = t;
    else atomicAdd(&out[blockIdx.x * 64 + threadIdx.x], t);
  }
}

template __global__ void colsum_vec_kernel<bf16>(const bf16*, float*,
                                                 long long, int, int);
template __global__ void colsum_vec_kernel<float>(const float*, float*,
                                                  long long, int, int);

template __global__ void colsum_kernel<bf16>(const bf16*, float*, long long,
                                             int, int);
template __global__ void colsum_kernel<float>(const float*, float*, long long,
                                              int, int);

void launch_colsum(bool is_bf16, const void* x, float* out, long long R, int C,
                   hipStream_t s) {
  const int cgrid = (C + 63) / 64;
  int target = 1024 / (cgrid > 0 ? cgrid : 1);
  if (target < 1) target = 1;
  long long rpb = (R + target - 1) / target;
  if (rpb < 128) rpb = 128;
  int gy = (int)((R + rpb - 1) / rpb);
  dim3 grid(cgrid, gy);
  if ((C % 64) == 0) {
    if (is_bf16)
      hipLaunchKernelGGL(colsum_vec_kernel<bf16>, grid, dim3(kBlock), 0, s,
                         (const bf16*)x, out, R, C, (int)rpb);
    else
      hipLaunchKernelGGL(colsum_vec_kernel<float>, grid, dim3(kBlock), 0, s,
                         (const float*)x, out, R, C, (int)rpb);
    return;
  }
  if (is_bf16)
    hipLaunchKernelGGL(colsum_kernel<bf16>, grid, dim3(kBlock), 0, s,
                       (const bf16*)x, out, R, C, (int)rpb);
  else
    hipLaunchKernelGGL(colsum_kernel<float>, grid, dim3(kBlock), 0, s,
                       (const float*)x, out, R, C, (int)rpb);
}
