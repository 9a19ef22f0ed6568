// Fused optimizer kernels (gfx950).
//
// The reference steps torch.optim.SGD per batch on CPU
// (/root/reference/demo.py:34,47). Here the whole step is one
// HBM-rate streaming kernel over a flat buffer (runtime/arena.py):
// p/g in fp32 or bf16, momentum/Adam moments always fp32.
// 16 B/lane vectorized (G13), grid-stride, no host sync.
#include "common.h"

// tail == true: scalar path for the last (n % kElems) elements.
// zero_after: write zeros back to g after consuming it — fuses the next
// iteration's zero_grad fill into this pass (one launch fewer per step).
template <typename T>
__global__ void sgd_kernel(T* __restrict__ p, T* __restrict__ g,
                           float* __restrict__ m, long long n, float lr,
                           float momentum, float weight_decay,
                           int zero_after) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  const bool has_m = (m != nullptr);
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT pv = reinterpret_cast<const typename VT::VecT*>(p)[i];
    typename VT::VecT gv = reinterpret_cast<const typename VT::VecT*>(g)[i];
    float pf[V], gf[V];
    VT::to_float(pv, pf);
    VT::to_float(gv, gf);
    if (has_m) {
      f32x4 mv[V / 4];
#pragma unroll
      for (int q = 0; q < V / 4; ++q)
        mv[q] = reinterpret_cast<const f32x4*>(m)[i * (V / 4) + q];
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float grad = fmaf(weight_decay, pf[k], gf[k]);
        float mom = fmaf(momentum, mv[k / 4][k % 4], grad);
        mv[k / 4][k % 4] = mom;
        pf[k] = fmaf(-lr, mom, pf[k]);
      }
#pragma unroll
      for (int q = 0; q < V / 4; ++q)
        reinterpret_cast<f32x4*>(m)[i * (V / 4) + q] = mv[q];
    } else {
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float grad = fmaf(weight_decay, pf[k], gf[k]);
        pf[k] = fmaf(-lr, grad, pf[k]);
      }
    }
    VT::from_float(pf, pv);
    reinterpret_cast<typename VT::VecT*>(p)[i] = pv;
    if (zero_after)
      reinterpret_cast<typename VT::VecT*>(g)[i] = typename VT::VecT{};
  }
  // scalar tail
  long long tail_start = nvec * V;
  for (long long i = tail_start + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    float pf = (float)p[i], gf = (float)g[i];
    float grad = fmaf(weight_decay, pf, gf);
    if (has_m) {
      float mom = fmaf(momentum, m[i], grad);
      m[i] = mom;
      grad = mom;
    }
    p[i] = (T)fmaf(-lr, grad, pf);
    if (zero_after) g[i] = (T)0.f;
  }
}

template <typename T>
__global__ void adam_kernel(T* __restrict__ p, T* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            long long n, float lr, float beta1, float beta2,
                            float eps, float weight_decay, float inv_bc1,
                            float inv_sqrt_bc2, int zero_after) {
  using VT = VecTraits<T>;
  constexpr int V = VT::kElems;
  const long long nvec = n / V;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long long)gridDim.x * blockDim.x) {
    typename VT::VecT pv = reinterpret_cast<const typename VT::VecT*>(p)[i];
    typename VT::VecT gv = reinterpret_cast<const typename VT::VecT*>(g)[i];
    float pf[V], gf[V];
    VT::to_float(pv, pf);
    VT::to_float(gv, gf);
    f32x4 mv[V / 4], vv[V / 4];
#pragma unroll
    for (int q = 0; q < V / 4; ++q) {
      mv[q] = reinterpret_cast<const f32x4*>(m)[i * (V / 4) + q];
      vv[q] = reinterpret_cast<const f32x4*>(v)[i * (V / 4) + q];
    }
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float grad = fmaf(weight_decay, pf[k], gf[k]);
      float m_new = fmaf(beta1, mv[k / 4][k % 4], (1.f - beta1) * grad);
      float v_new = fmaf(beta2, vv[k / 4][k % 4], (1.f - beta2) * grad * grad);
      mv[k / 4][k % 4] = m_new;
      vv[k / 4][k % 4] = v_new;
      // p -= lr * (m/bc1) / (sqrt(v/bc2) + eps)
      float denom = fmaf(sqrtf(v_new), inv_sqrt_bc2, eps);
      pf[k] = fmaf(-lr * inv_bc1, m_new / denom, pf[k]);
    }
#pragma unroll
    for (int q = 0; q < V / 4; ++q) {
      reinterpret_cast<f32x4*>(m)[i * (V / 4) + q] = mv[q];
      reinterpret_cast<f32x4*>(v)[i * (V / 4) + q] = vv[q];
    }
    VT::from_float(pf, pv);
    reinterpret_cast<typename VT::VecT*>(p)[i] = pv;
    if (zero_after)
      reinterpret_cast<typename VT::VecT*>(g)[i] = typename VT::VecT{};
  }
  long long tail_start = nvec * V;
  for (long long i = tail_start + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    float pf = (float)p[i], gf = (float)g[i];
    float grad = fmaf(weight_decay, pf, gf);
    float m_new = fmaf(beta1, m[i], (1.f - beta1) * grad);
    float v_new = fmaf(beta2, v[i], (1.f - beta2) * grad * grad);
    m[i] = m_new;
    v[i] = v_new;
    float denom = fmaf(sqrtf(v_new), inv_sqrt_bc2, eps);
    p[i] = (T)fmaf(-lr * inv_bc1, m_new / denom, pf);
    if (zero_after) g[i] = (T)0.f;
  }
}

// explicit instantiations referenced from bindings.cpp
template __global__ void sgd_kernel<float>(float*, float*, float*, long long,
                                           float, float, float, int);
template __global__ void sgd_kernel<bf16>(bf16*, bf16*, float*, long long,
                                          float, float, float, int);
template __global__ void adam_kernel<float>(float*, float*, float*, float*,
                                            long long, float, float, float,
                                            float, float, float, float, int);
template __global__ void adam_kernel<bf16>(bf16*, bf16*, float*, float*,
                                           long long, float, float, float, float,
                                           float, float, float, int);

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_sgd(bool is_bf16, void* p, void* g, float* m, long long n,
                float lr, float momentum, float wd, int zero_after,
                hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(sgd_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (bf16*)p, (bf16*)g, m, n, lr, momentum, wd, zero_after);
  else
    hipLaunchKernelGGL(sgd_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (float*)p, (float*)g, m, n, lr, momentum, wd,
                       zero_after);
}

void launch_adam(bool is_bf16, void* p, void* g, float* m, float* v,
                 long long n, float lr, float b1, float b2, float eps, float wd,
                 float inv_bc1, float inv_sqrt_bc2, int zero_after,
                 hipStream_t s) {
  const int grid = elementwise_grid(n / 8 + 1);
  if (is_bf16)
    hipLaunchKernelGGL(adam_kernel<bf16>, dim3(grid), dim3(kBlock), 0, s,
                       (bf16*)p, (bf16*)g, m, v, n, lr, b1, b2, eps, wd,
                       inv_bc1, inv_sqrt_bc2, zero_after);
  else
    hipLaunchKernelGGL(adam_kernel<float>, dim3(grid), dim3(kBlock), 0, s,
                       (float*)p, (float*)g, m, v, n, lr, b1, b2, eps, wd,
                       inv_bc1, inv_sqrt_bc2, zero_after);
}
