// Elementwise kernels (gfx950): ReLU, residual add+ReLU, GELU — the glue
// ops of the model zoo, vectorized to 16 B/lane (G13), grid-stride.
#include "common.h"

template <typename T>
__global__ void relu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT xv = reinterpret_cast<const typename VT::VecT*>(x)[i];
    float f[V];
    VT::to_float(xv, f);
#pragma unroll
    for (int k = 0; k < V; ++k) f[k] = fmaxf(f[k], 0.f);
    typename VT::VecT ov;
    VT::from_float(f, ov);
    reinterpret_cast<typename VT::VecT*>(y)[i] = ov;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    y[i] = (T)fmaxf((float)x[i], 0.f);
}

// dx = dy * (y > 0)
template <typename T>
__global__ void relu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ y,
                                T* __restrict__ dx, long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT dv = reinterpret_cast<const typename VT::VecT*>(dy)[i];
    typename VT::VecT yv = reinterpret_cast<const typename VT::VecT*>(y)[i];
    float df[V], yf[V];
    VT::to_float(dv, df);
    VT::to_float(yv, yf);
#pragma unroll
    for (int k = 0; k < V; ++k) df[k] = yf[k] > 0.f ? df[k] : 0.f;
    typename VT::VecT ov;
    VT::from_float(df, ov);
    reinterpret_cast<typename VT::VecT*>(dx)[i] = ov;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    dx[i] = (T)((float)y[i] > 0.f ? (float)dy[i] : 0.f);
}

// y = relu(a + b) — the ResNet residual join, fused
template <typename T>
__global__ void add_relu_fwd_kernel(const T* __restrict__ a,
                                    const T* __restrict__ b, T* __restrict__ y,
                                    long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT av = reinterpret_cast<const typename VT::VecT*>(a)[i];
    typename VT::VecT bv = reinterpret_cast<const typename VT::VecT*>(b)[i];
    float af[V], bf[V];
    VT::to_float(av, af);
    VT::to_float(bv, bf);
#pragma unroll
    for (int k = 0; k < V; ++k) af[k] = fmaxf(af[k] + bf[k], 0.f);
    typename VT::VecT ov;
    VT::from_float(af, ov);
    reinterpret_cast<typename VT::VecT*>(y)[i] = ov;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    y[i] = (T)fmaxf((float)a[i] + (float)b[i], 0.f);
}

// z = a + alpha * b — the LoRA combine (base + scaling * delta), fused:
// one kernel instead of a scale pass plus an add pass (3 tensor passes
// instead of 5 at ~6 TB/s HBM-bound)
template <typename T>
__global__ void add_scaled_fwd_kernel(const T* __restrict__ a,
                                      const T* __restrict__ b,
                                      T* __restrict__ z, float alpha,
                                      long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT av = reinterpret_cast<const typename VT::VecT*>(a)[i];
    typename VT::VecT bv = reinterpret_cast<const typename VT::VecT*>(b)[i];
    float af[V], bf[V];
    VT::to_float(av, af);
    VT::to_float(bv, bf);
#pragma unroll
    for (int k = 0; k < V; ++k) af[k] = fmaf(alpha, bf[k], af[k]);
    typename VT::VecT ov;
    VT::from_float(af, ov);
    reinterpret_cast<typename VT::VecT*>(z)[i] = ov;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    z[i] = (T)fmaf(alpha, (float)b[i], (float)a[i]);
}

// z = alpha * x (the add_scaled backward for the delta operand)
template <typename T>
__global__ void scale_fwd_kernel(const T* __restrict__ x, T* __restrict__ z,
                                 float alpha, long long n) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT xv = reinterpret_cast<const typename VT::VecT*>(x)[i];
    float f[V];
    VT::to_float(xv, f);
#pragma unroll
    for (int k = 0; k < V; ++k) f[k] *= alpha;
    typename VT::VecT ov;
    VT::from_float(f, ov);
    reinterpret_cast<typename VT::VecT*>(z)[i] = ov;
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    z[i] = (T)(alpha * (float)x[i]);
}

// tanh-approx GELU (BERT): y = 0.5x(1+tanh(0.79788456(x+0.044715x^3)))
DEVINL float gelu_of(float v) {
  float inner = 0.7978845608f * fmaf(0.044715f * v * v, v, v);
  return 0.5f * v * (1.f + tanhf(inner));
}

template <typename T>
__global__ void gelu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                long long n) {
  constexpr int V = VecTraits<T>::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    float f[V];
    vload16(x + i * V, f);
#pragma unroll
    for (int q = 0; q < V; ++q) f[q] = gelu_of(f[q]);
    vstore16(y + i * V, f);
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    y[i] = (T)gelu_of((float)x[i]);
}

DEVINL float gelu_grad(float v) {
  float inner = 0.7978845608f * fmaf(0.044715f * v * v, v, v);
  float t = tanhf(inner);
  float dinner = 0.7978845608f * fmaf(3.f * 0.044715f * v, v, 1.f);
  return 0.5f * (1.f + t) + 0.5f * v * (1.f - t * t) * dinner;
}

template <typename T>
__global__ void gelu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                T* __restrict__ dx, long long n) {
  constexpr int V = VecTraits<T>::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    float fx[V], fd[V];
    vload16(x + i * V, fx);
    vload16(dy + i * V, fd);
#pragma unroll
    for (int q = 0; q < V; ++q) fd[q] *= gelu_grad(fx[q]);
    vstore16(dx + i * V, fd);
  }
  for (long long i = nvec * V + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    dx[i] = (T)((float)dy[i] * gelu_grad((float)x[i]));
}

#define INST_EW(T)                                                           \
  template __global__ void relu_fwd_kernel<T>(const T*, T*, long long);      \
  template __global__ void relu_bwd_kernel<T>(const T*, const T*, T*,        \
                                              long long);                    \
  template __global__ void add_relu_fwd_kernel<T>(const T*, const T*, T*,    \
                                                  long long);                \
  template __global__ void add_scaled_fwd_kernel<T>(const T*, const T*, T*,  \
                                                    float, long long);       \
  template __global__ void scale_fwd_kernel<T>(const T*, T*, float,          \
                                               long long);                   \
  template __global__ void gelu_fwd_kernel<T>(const T*, T*, long long);      \
  template __global__ void gelu_bwd_kernel<T>(const T*, const T*, T*,        \
                                              long long);

INST_EW(float)
INST_EW(bf16)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_relu_fwd(bool is_bf16, const void* x, void* y, long long n,
                     hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(relu_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (const bf16*)x, (bf16*)y, n);
  else
    hipLaunchKernelGGL(relu_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (const float*)x, (float*)y, n);
}

void launch_relu_bwd(bool is_bf16, const void* dy, const void* y, void* dx,
                     long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(relu_bwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (const bf16*)dy, (const bf16*)y, (bf16*)dx, n);
  else
    hipLaunchKernelGGL(relu_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (const float*)dy, (const float*)y, (float*)dx, n);
}

void launch_add_relu_fwd(bool is_bf16, const void* a, const void* b, void* y,
                         long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(add_relu_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0,
                       s, (const bf16*)a, (const bf16*)b, (bf16*)y, n);
  else
    hipLaunchKernelGGL(add_relu_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       s, (const float*)a, (const float*)b, (float*)y, n);
}

void launch_add_scaled_fwd(bool is_bf16, const void* a, const void* b, void* z,
                           float alpha, long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(add_scaled_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0,
                       s, (const bf16*)a, (const bf16*)b, (bf16*)z, alpha, n);
  else
    hipLaunchKernelGGL(add_scaled_fwd_kernel<float>, dim3(grid), dim3(kBlock),
                       0, s, (const float*)a, (const float*)b, (float*)z, alpha,
                       n);
}

void launch_scale_fwd(bool is_bf16, const void* x, void* z, float alpha,
                      long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(scale_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (const bf16*)x, (bf16*)z, alpha, n);
  else
    hipLaunchKernelGGL(scale_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (const float*)x, (float*)z, alpha, n);
}

void launch_gelu_fwd(bool is_bf16, const void* x, void* y, long long n,
                     hipStream_t s) {
  const int grid = elementwise_grid(n / 4 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(gelu_fwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (const bf16*)x, (bf16*)y, n);
  else
    hipLaunchKernelGGL(gelu_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (const float*)x, (float*)y, n);
}

void launch_gelu_bwd(bool is_bf16, const void* dy, const void* x, void* dx,
                     long long n, hipStream_t s) {
  const int grid = elementwise_grid(n / 4 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(gelu_bwd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (const bf16*)dy, (const bf16*)x, (bf16*)dx, n);
  else
    hipLaunchKernelGGL(gelu_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (const float*)dy, (const float*)x, (float*)dx, n);
}
