// Shared device helpers for the baton_amd gfx950 kernels.
//
// Written for CDNA4 (MI355X) only: 64-wide wavefronts, 256 CUs / 8 XCDs,
// HBM3E at ~8 TB/s. Every memory-bound kernel vectorizes to 16 B/lane
// (G13 in the CDNA HIP guide: hipcc does not auto-vectorize bf16 loads).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEVINL __device__ __forceinline__

constexpr int kWave = 64;           // CDNA wavefront width (not 32)
constexpr int kBlock = 256;         // default block: 4 waves
// Memory-bound grid cap: 256 CUs x 8 blocks, grid-stride the rest (G11).
constexpr int kMaxGrid = 2048;

static inline int elementwise_grid(long long n_vec, int block = kBlock) {
  long long b = (n_vec + block - 1) / block;
  if (b > kMaxGrid) b = kMaxGrid;
  if (b < 1) b = 1;
  return (int)b;
}

#include "gemm_strides.h"

// ---- dtype traits: 16-byte vectors ---------------------------------------

using bf16 = __hip_bfloat16;
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

template <typename T>
struct VecTraits;

template <>
struct VecTraits<float> {
  static constexpr int kElems = 4;  // 16 B
  using VecT = f32x4;
  DEVINL static void to_float(const VecT& v, float* out) {
#pragma unroll
    for (int i = 0; i < 4; ++i) out[i] = v[i];
  }
  DEVINL static void from_float(const float* in, VecT& v) {
#pragma unroll
    for (int i = 0; i < 4; ++i) v[i] = in[i];
  }
};

template <>
struct VecTraits<bf16> {
  static constexpr int kElems = 8;  // 16 B
  using VecT = s16x8;
  DEVINL static void to_float(const VecT& v, float* out) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      unsigned int u = ((unsigned int)(unsigned short)v[i]) << 16;
      out[i] = __uint_as_float(u);
    }
  }
  DEVINL static void from_float(const float* in, VecT& v) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      // round-to-nearest-even f32 -> bf16
      unsigned int u = __float_as_uint(in[i]);
      unsigned int rounded = u + 0x7fff + ((u >> 16) & 1);
      v[i] = (short)(rounded >> 16);
    }
  }
};

DEVINL float bf16_to_f32(unsigned short h) {
  return __uint_as_float(((unsigned int)h) << 16);
}
DEVINL unsigned short f32_to_bf16(float f) {
  unsigned int u = __float_as_uint(f);
  unsigned int rounded = u + 0x7fff + ((u >> 16) & 1);
  return (unsigned short)(rounded >> 16);
}

// 16-B vector load/store to/from fp32 lanes (G13: hipcc does not
// auto-vectorize bf16 element loops)
template <typename T>
DEVINL void vload16(const T* p, float* f) {
  using VT = VecTraits<T>;
  typename VT::VecT v = *reinterpret_cast<const typename VT::VecT*>(p);
  VT::to_float(v, f);
}
template <typename T>
DEVINL void vstore16(T* p, const float* f) {
  using VT = VecTraits<T>;
  typename VT::VecT v;
  VT::from_float(f, v);
  *reinterpret_cast<typename VT::VecT*>(p) = v;
}

// ---- reductions ----------------------------------------------------------

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, kWave);
  return v;  // valid in lane 0 of the wave
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, kWave));
  return v;
}

// Block-level sum reduction; returns the total in every thread.
// Requires a __shared__ float scratch[block/kWave] from the caller.
DEVINL float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int nwaves = blockDim.x / kWave;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

// ---- canonical K-slice LDS image (shared by gemm.hip / conv.hip) ---------
// bf16: LINEAR 64-B rows (32 elements) with a 16-B slot XOR swizzle —
// compatible with global_load_lds (lane-linear dest) and bank-conflict-free
// for both ds_read_b128 fragment groups and transposed scalar writes
// (contiguous k elements spread banks naturally).
// f32: padded rows (+2 elements) with scalar b32 fragment reads.
constexpr int kBKElems = 32;   // K-slice width in elements

template <typename T>
constexpr int lds_row_elems() {
  return sizeof(T) == 2 ? kBKElems : kBKElems + 2;
}

template <typename T>
DEVINL int lds_off(int row, int col) {
  if constexpr (sizeof(T) == 2) {
    const int sl = col >> 3;
    return row * kBKElems + ((sl ^ ((row >> 2) & 3)) << 3) + (col & 7);
  } else {
    return row * (kBKElems + 2) + col;
  }
}
