// Implicit-GEMM convolution (gfx950), NHWC layout, forward + dgrad + wgrad.
//
// Required by the ResNet federated configs (SURVEY.md §2.2 row
// "Conv/BatchNorm"; the reference has no conv at all). Design: the
// convolution IS a GEMM on MFMA — no im2col materialization; the A/B tiles
// are gathered straight from NHWC tensors into LDS:
//
//   forward  y[p, co]  = sum_k x_gather(p, k) * w[co, k]          k=(kh,kw,ci)
//   dgrad    dx[q, ci] = sum_k dy_gather(q, k) * w_t(k, ci)       k=(kh,kw,co)
//   wgrad    dw[co, r] = sum_p dy[p, co] * x_gather2(p, r)        r=(kh,kw,ci)
//
// NHWC makes every innermost gather run contiguous in channels, so when the
// channel count is a multiple of 32 (every ResNet layer except conv1) the
// K-slice of a tile sits inside one (kh,kw) tap and stages with 16-B
// vector loads; otherwise a scalar-gather fallback handles ragged shapes.
//
// Performance structure:
//   * tiles are TEMPLATED: 128x128 (2x2 waves, 64x64 each) for wide
//     layers, 128x64 (4x1 waves, 32x64 each) when the output channel dim
//     is <= 64 (ResNet layer1/stem would otherwise idle half the waves),
//     64x128 (1x4 waves) for wgrad of 64-filter layers;
//   * DOUBLE-BUFFERED LDS: the k-tile t+1 is staged (global->reg->ds_write)
//     while MFMAs consume tile t — one barrier per k-step (guide §5.5
//     minimum-2-phase recipe);
//   * wgrad splits the pixel contraction over grid.z (fp32 atomic
//     accumulation) so its tiny tile grid still fills 256 CUs;
//   * LDS rows padded +8 bf16 so ds_read_b128 fragment groups are
//     bank-conflict-free (Guideline 4).
#include "common.h"

constexpr int CBK = 32;
constexpr int CBKP = CBK + 8;
constexpr int CFRAG = 16;

struct ConvShape {
  int N, H, W, Cin, Cout, KH, KW, stride, pad, HO, WO;
};

// ---- staging ---------------------------------------------------------------

// A tile (forward): rows = output pixels, cols = k (kh,kw,ci).
template <typename T, int ROWS>
DEVINL void stage_fwd_A(T* __restrict__ lds, const T* __restrict__ x,
                        const ConvShape sh, int m0, int k0, int Mtot, int Ktot) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int THREADS_PER_ROW = CBK / ELEMS;
  constexpr int ROWS_PER_PASS = kBlock / THREADS_PER_ROW;
  using VT = typename VecTraits<T>::VecT;
  const bool fast = (sh.Cin % CBK) == 0;
#pragma unroll
  for (int p = 0; p < ROWS / ROWS_PER_PASS; ++p) {
    int idx = p * kBlock + threadIdx.x;
    int row = idx / THREADS_PER_ROW;
    int kc = (idx % THREADS_PER_ROW) * ELEMS;
    int m = m0 + row;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    int wo = m % sh.WO, tmp = m / sh.WO;
    int ho = tmp % sh.HO, n = tmp / sh.HO;
    if (fast && m < Mtot) {
      int k = k0 + kc;
      int ci = k % sh.Cin, tap = k / sh.Cin;
      int kw = tap % sh.KW, kh = tap / sh.KW;
      int hi = ho * sh.stride - sh.pad + kh;
      int wi = wo * sh.stride - sh.pad + kw;
      if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W) {
        v = *reinterpret_cast<const VT*>(
            &x[(((long long)n * sh.H + hi) * sh.W + wi) * sh.Cin + ci]);
      } else {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j) {
        int k = k0 + kc + j;
        vp[j] = (T)0.f;
        if (m < Mtot && k < Ktot) {
          int ci = k % sh.Cin, tap = k / sh.Cin;
          int kw = tap % sh.KW, kh = tap / sh.KW;
          int hi = ho * sh.stride - sh.pad + kh;
          int wi = wo * sh.stride - sh.pad + kw;
          if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W)
            vp[j] = x[(((long long)n * sh.H + hi) * sh.W + wi) * sh.Cin + ci];
        }
      }
    }
    *reinterpret_cast<VT*>(&lds[row * CBKP + kc]) = v;
  }
}

// B tile (forward): rows = Cout; w is [Cout][KH*KW*Cin] row-major: direct.
template <typename T, int ROWS>
DEVINL void stage_fwd_B(T* __restrict__ lds, const T* __restrict__ w,
                        int n0, int k0, int Ntot, int Ktot) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int THREADS_PER_ROW = CBK / ELEMS;
  constexpr int ROWS_PER_PASS = kBlock / THREADS_PER_ROW;
  using VT = typename VecTraits<T>::VecT;
#pragma unroll
  for (int p = 0; p < ROWS / ROWS_PER_PASS; ++p) {
    int idx = p * kBlock + threadIdx.x;
    int row = idx / THREADS_PER_ROW;
    int kc = (idx % THREADS_PER_ROW) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    if (n0 + row < Ntot && k0 + kc + ELEMS <= Ktot) {
      v = *reinterpret_cast<const VT*>(&w[(long long)(n0 + row) * Ktot + k0 + kc]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        vp[j] = (n0 + row < Ntot && k0 + kc + j < Ktot)
                    ? w[(long long)(n0 + row) * Ktot + k0 + kc + j]
                    : (T)0.f;
    }
    *reinterpret_cast<VT*>(&lds[row * CBKP + kc]) = v;
  }
}

// A tile (dgrad): rows = input pixels q=(n,h,w); k = (kh,kw,co).
template <typename T, int ROWS>
DEVINL void stage_dgrad_A(T* __restrict__ lds, const T* __restrict__ dy,
                          const ConvShape sh, int m0, int k0, int Mtot) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int THREADS_PER_ROW = CBK / ELEMS;
  constexpr int ROWS_PER_PASS = kBlock / THREADS_PER_ROW;
  using VT = typename VecTraits<T>::VecT;
  const bool fast = (sh.Cout % CBK) == 0;
#pragma unroll
  for (int p = 0; p < ROWS / ROWS_PER_PASS; ++p) {
    int idx = p * kBlock + threadIdx.x;
    int row = idx / THREADS_PER_ROW;
    int kc = (idx % THREADS_PER_ROW) * ELEMS;
    int q = m0 + row;
    int wq = q % sh.W, tmp = q / sh.W;
    int hq = tmp % sh.H, n = tmp / sh.H;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
    if (q < Mtot) {
      if (fast) {
        int k = k0 + kc;
        int co = k % sh.Cout, tap = k / sh.Cout;
        int kw = tap % sh.KW, kh = tap / sh.KW;
        int hnum = hq + sh.pad - kh, wnum = wq + sh.pad - kw;
        if (hnum >= 0 && wnum >= 0 && hnum % sh.stride == 0 &&
            wnum % sh.stride == 0) {
          int ho = hnum / sh.stride, wo = wnum / sh.stride;
          if (ho < sh.HO && wo < sh.WO)
            v = *reinterpret_cast<const VT*>(
                &dy[(((long long)n * sh.HO + ho) * sh.WO + wo) * sh.Cout + co]);
        }
      } else {
        const int Ktot = sh.KH * sh.KW * sh.Cout;
#pragma unroll
        for (int j = 0; j < ELEMS; ++j) {
          int k = k0 + kc + j;
          if (k < Ktot) {
            int co = k % sh.Cout, tap = k / sh.Cout;
            int kw = tap % sh.KW, kh = tap / sh.KW;
            int hnum = hq + sh.pad - kh, wnum = wq + sh.pad - kw;
            if (hnum >= 0 && wnum >= 0 && hnum % sh.stride == 0 &&
                wnum % sh.stride == 0) {
              int ho = hnum / sh.stride, wo = wnum / sh.stride;
              if (ho < sh.HO && wo < sh.WO)
                vp[j] = dy[(((long long)n * sh.HO + ho) * sh.WO + wo) * sh.Cout + co];
            }
          }
        }
      }
    }
    *reinterpret_cast<VT*>(&lds[row * CBKP + kc]) = v;
  }
}

// B tile (dgrad): rows = ci; k=(kh,kw,co); w[co][kh][kw][ci] ci-contiguous
// => transposed staging (8 ci per vector load, 8 LDS rows).
template <typename T, int ROWS>
DEVINL void stage_dgrad_B(T* __restrict__ lds, const T* __restrict__ w,
                          const ConvShape sh, int n0, int k0) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int VECS_PER_K = ROWS / ELEMS;
  using VT = typename VecTraits<T>::VecT;
  constexpr int TOTAL = CBK * VECS_PER_K;
  const int Ktot = sh.KH * sh.KW * sh.Cout;
#pragma unroll
  for (int p = 0; p < (TOTAL + kBlock - 1) / kBlock; ++p) {
    int idx = p * kBlock + threadIdx.x;
    if (idx >= TOTAL) break;
    int k = idx % CBK;
    int r = (idx / CBK) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    int kk = k0 + k;
    int ci = n0 + r;
    if (kk < Ktot && ci + ELEMS <= sh.Cin) {
      int co = kk % sh.Cout, tap = kk / sh.Cout;
      int kw = tap % sh.KW, kh = tap / sh.KW;
      v = *reinterpret_cast<const VT*>(
          &w[(((long long)co * sh.KH + kh) * sh.KW + kw) * sh.Cin + ci]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j) {
        vp[j] = (T)0.f;
        if (kk < Ktot && ci + j < sh.Cin) {
          int co = kk % sh.Cout, tap = kk / sh.Cout;
          int kw = tap % sh.KW, kh = tap / sh.KW;
          vp[j] = w[(((long long)co * sh.KH + kh) * sh.KW + kw) * sh.Cin + ci + j];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) lds[(r + j) * CBKP + k] = vp[j];
  }
}

// wgrad A: rows = co (transposed from dy [P, Cout]); k = pixel.
template <typename T, int ROWS>
DEVINL void stage_wgrad_A(T* __restrict__ lds, const T* __restrict__ dy,
                          const ConvShape sh, int m0, long long p0,
                          long long p_limit) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int VECS_PER_K = ROWS / ELEMS;
  using VT = typename VecTraits<T>::VecT;
  constexpr int TOTAL = CBK * VECS_PER_K;
#pragma unroll
  for (int pp = 0; pp < (TOTAL + kBlock - 1) / kBlock; ++pp) {
    int idx = pp * kBlock + threadIdx.x;
    if (idx >= TOTAL) break;
    int k = idx % CBK;
    int r = (idx / CBK) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    long long p = p0 + k;
    int co = m0 + r;
    if (p < p_limit && co + ELEMS <= sh.Cout) {
      v = *reinterpret_cast<const VT*>(&dy[p * sh.Cout + co]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        vp[j] = (p < p_limit && co + j < sh.Cout) ? dy[p * sh.Cout + co + j]
                                                  : (T)0.f;
    }
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) lds[(r + j) * CBKP + k] = vp[j];
  }
}

// wgrad B: rows = (kh,kw,ci) gathered from x; k = pixel.
template <typename T, int ROWS>
DEVINL void stage_wgrad_B(T* __restrict__ lds, const T* __restrict__ x,
                          const ConvShape sh, int n0, long long p0,
                          long long p_limit) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int VECS_PER_K = ROWS / ELEMS;
  using VT = typename VecTraits<T>::VecT;
  constexpr int TOTAL = CBK * VECS_PER_K;
  const int Rtot = sh.KH * sh.KW * sh.Cin;
#pragma unroll
  for (int pp = 0; pp < (TOTAL + kBlock - 1) / kBlock; ++pp) {
    int idx = pp * kBlock + threadIdx.x;
    if (idx >= TOTAL) break;
    int k = idx % CBK;
    int r = (idx / CBK) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
    long long p = p0 + k;
    int rr = n0 + r;
    if (p < p_limit && rr < Rtot) {
      int wo = (int)(p % sh.WO);
      long long t = p / sh.WO;
      int ho = (int)(t % sh.HO), n = (int)(t / sh.HO);
      const bool one_tap = (rr / sh.Cin) == ((rr + ELEMS - 1) / sh.Cin);
      if (one_tap) {
        int ci = rr % sh.Cin, tap = rr / sh.Cin;
        int kw = tap % sh.KW, kh = tap / sh.KW;
        int hi = ho * sh.stride - sh.pad + kh;
        int wi = wo * sh.stride - sh.pad + kw;
        if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W)
          v = *reinterpret_cast<const VT*>(
              &x[(((long long)n * sh.H + hi) * sh.W + wi) * sh.Cin + ci]);
      } else {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j) {
          int rj = rr + j;
          if (rj < Rtot) {
            int cij = rj % sh.Cin, tapj = rj / sh.Cin;
            int kwj = tapj % sh.KW, khj = tapj / sh.KW;
            int hij = ho * sh.stride - sh.pad + khj;
            int wij = wo * sh.stride - sh.pad + kwj;
            if (hij >= 0 && hij < sh.H && wij >= 0 && wij < sh.W)
              vp[j] = x[(((long long)n * sh.H + hij) * sh.W + wij) * sh.Cin + cij];
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) lds[(r + j) * CBKP + k] = vp[j];
  }
}

// ---- MFMA compute ----------------------------------------------------------

template <typename T, int MF, int NF>
DEVINL void conv_mma(const T* a_lds, const T* b_lds, f32x4 (&acc)[MF][NF],
                     int lane, int wm0, int wn0) {
  if constexpr (sizeof(T) == 2) {
    s16x8 a_frag[MF], b_frag[NF];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &a_lds[(wm0 + mf * CFRAG + (lane & 15)) * CBKP + (lane >> 4) * 8]);
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &b_lds[(wn0 + nf * CFRAG + (lane & 15)) * CBKP + (lane >> 4) * 8]);
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
  } else {
#pragma unroll
    for (int kk = 0; kk < CBK / 4; ++kk) {
      float a_s[MF], b_s[NF];
      const int kidx = kk * 4 + (lane >> 4);
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
        a_s[mf] = ((const float*)a_lds)[(wm0 + mf * CFRAG + (lane & 15)) * CBKP + kidx];
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        b_s[nf] = ((const float*)b_lds)[(wn0 + nf * CFRAG + (lane & 15)) * CBKP + kidx];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_s[mf], b_s[nf], acc[mf][nf], 0, 0, 0);
    }
  }
}

// ---- kernels (double-buffered LDS, one barrier per k-step) -----------------

template <typename T, int BM, int BN, int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(kBlock) void conv_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ y,
    ConvShape sh) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * CBKP];
  __shared__ T b_lds[2][BN * CBKP];
  const int Mtot = sh.N * sh.HO * sh.WO;
  const int Ntot = sh.Cout;
  const int Ktot = sh.KH * sh.KW * sh.Cin;
  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM, wn0 = (wid % WAVES_N) * WN;
  f32x4 acc[MF][NF] = {};
  const int nk = (Ktot + CBK - 1) / CBK;
  stage_fwd_A<T, BM>(a_lds[0], x, sh, m0, 0, Mtot, Ktot);
  stage_fwd_B<T, BN>(b_lds[0], w, n0, 0, Ntot, Ktot);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      stage_fwd_A<T, BM>(a_lds[cur ^ 1], x, sh, m0, (kt + 1) * CBK, Mtot, Ktot);
      stage_fwd_B<T, BN>(b_lds[cur ^ 1], w, n0, (kt + 1) * CBK, Ntot, Ktot);
    }
    conv_mma<T, MF, NF>(a_lds[cur], b_lds[cur], acc, lane, wm0, wn0);
    __syncthreads();
  }
  const int col_in_frag = lane & 15, row_base = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MF; ++mf)
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm0 + mf * CFRAG + row_base + r;
        int col = n0 + wn0 + nf * CFRAG + col_in_frag;
        if (row < Mtot && col < Ntot)
          y[(long long)row * Ntot + col] = (T)acc[mf][nf][r];
      }
}

template <typename T, int BM, int BN, int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(kBlock) void conv_dgrad_kernel(
    const T* __restrict__ dy, const T* __restrict__ w, T* __restrict__ dx,
    ConvShape sh) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * CBKP];
  __shared__ T b_lds[2][BN * CBKP];
  const int Mtot = sh.N * sh.H * sh.W;
  const int Ntot = sh.Cin;
  const int Ktot = sh.KH * sh.KW * sh.Cout;
  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM, wn0 = (wid % WAVES_N) * WN;
  f32x4 acc[MF][NF] = {};
  const int nk = (Ktot + CBK - 1) / CBK;
  stage_dgrad_A<T, BM>(a_lds[0], dy, sh, m0, 0, Mtot);
  stage_dgrad_B<T, BN>(b_lds[0], w, sh, n0, 0);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      stage_dgrad_A<T, BM>(a_lds[cur ^ 1], dy, sh, m0, (kt + 1) * CBK, Mtot);
      stage_dgrad_B<T, BN>(b_lds[cur ^ 1], w, sh, n0, (kt + 1) * CBK);
    }
    conv_mma<T, MF, NF>(a_lds[cur], b_lds[cur], acc, lane, wm0, wn0);
    __syncthreads();
  }
  const int col_in_frag = lane & 15, row_base = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MF; ++mf)
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm0 + mf * CFRAG + row_base + r;
        int col = n0 + wn0 + nf * CFRAG + col_in_frag;
        if (row < Mtot && col < Ntot)
          dx[(long long)row * Ntot + col] = (T)acc[mf][nf][r];
      }
}

// Split-K over pixels: the wgrad output tile grid is tiny (e.g. ResNet
// 3x3x64x64 -> a handful of tiles) while the contraction runs over all
// output pixels, so grid.z slices the pixel range; partials land in the
// fp32 output via atomicAdd (94% of step time before this fix).
template <typename T, int BM, int BN, int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(kBlock) void conv_wgrad_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, float* __restrict__ dw,
    ConvShape sh, long long p_chunk) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * CBKP];
  __shared__ T b_lds[2][BN * CBKP];
  const int Mtot = sh.Cout;
  const int Ntot = sh.KH * sh.KW * sh.Cin;
  const long long Ptot = (long long)sh.N * sh.HO * sh.WO;
  const long long p_begin = (long long)blockIdx.z * p_chunk;
  const long long p_end = min(p_begin + p_chunk, Ptot);
  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM, wn0 = (wid % WAVES_N) * WN;
  f32x4 acc[MF][NF] = {};
  const long long nk = (p_end - p_begin + CBK - 1) / CBK;
  if (nk <= 0) return;
  stage_wgrad_A<T, BM>(a_lds[0], dy, sh, m0, p_begin, p_end);
  stage_wgrad_B<T, BN>(b_lds[0], x, sh, n0, p_begin, p_end);
  __syncthreads();
  for (long long kt = 0; kt < nk; ++kt) {
    const int cur = (int)(kt & 1);
    if (kt + 1 < nk) {
      stage_wgrad_A<T, BM>(a_lds[cur ^ 1], dy, sh, m0,
                           p_begin + (kt + 1) * CBK, p_end);
      stage_wgrad_B<T, BN>(b_lds[cur ^ 1], x, sh, n0,
                           p_begin + (kt + 1) * CBK, p_end);
    }
    conv_mma<T, MF, NF>(a_lds[cur], b_lds[cur], acc, lane, wm0, wn0);
    __syncthreads();
  }
  const int col_in_frag = lane & 15, row_base = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MF; ++mf)
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm0 + mf * CFRAG + row_base + r;
        int col = n0 + wn0 + nf * CFRAG + col_in_frag;
        if (row < Mtot && col < Ntot) {
          if (gridDim.z == 1)
            dw[(long long)row * Ntot + col] = acc[mf][nf][r];
          else
            atomicAdd(&dw[(long long)row * Ntot + col], acc[mf][nf][r]);
        }
      }
}

// Instantiations: Big = 128x128 (2x2), NarrowN = 128x64 (4x1),
// NarrowM (wgrad) = 64x128 (1x4).
#define INST_CONV(T)                                                        \
  template __global__ void conv_fwd_kernel<T, 128, 128, 2, 2>(              \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_fwd_kernel<T, 128, 64, 4, 1>(               \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 128, 2, 2>(            \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 64, 4, 1>(             \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_wgrad_kernel<T, 128, 128, 2, 2>(            \
      const T*, const T*, float*, ConvShape, long long);                    \
  template __global__ void conv_wgrad_kernel<T, 64, 128, 1, 4>(             \
      const T*, const T*, float*, ConvShape, long long);

INST_CONV(bf16)
INST_CONV(float)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

static ConvShape make_shape(int N, int H, int W, int Cin, int Cout, int KH,
                            int KW, int stride, int pad) {
  ConvShape sh;
  sh.N = N; sh.H = H; sh.W = W; sh.Cin = Cin; sh.Cout = Cout;
  sh.KH = KH; sh.KW = KW; sh.stride = stride; sh.pad = pad;
  sh.HO = (H + 2 * pad - KH) / stride + 1;
  sh.WO = (W + 2 * pad - KW) / stride + 1;
  return sh;
}

void launch_conv_fwd(bool is_bf16, const void* x, const void* w, void* y,
                     int N, int H, int W, int Cin, int Cout, int KH, int KW,
                     int stride, int pad, hipStream_t s) {
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  long long M = (long long)N * sh.HO * sh.WO;
  const bool narrow = Cout <= 64;
  const int BN_ = narrow ? 64 : 128;
  dim3 grid((Cout + BN_ - 1) / BN_, (M + 127) / 128);
  #define FWD(T, BM, BN, WM, WN)                                            \
    hipLaunchKernelGGL((conv_fwd_kernel<T, BM, BN, WM, WN>), grid,          \
                       dim3(kBlock), 0, s, (const T*)x, (const T*)w, (T*)y, sh)
  if (is_bf16) { if (narrow) FWD(bf16, 128, 64, 4, 1); else FWD(bf16, 128, 128, 2, 2); }
  else { if (narrow) FWD(float, 128, 64, 4, 1); else FWD(float, 128, 128, 2, 2); }
  #undef FWD
}

void launch_conv_dgrad(bool is_bf16, const void* dy, const void* w, void* dx,
                       int N, int H, int W, int Cin, int Cout, int KH, int KW,
                       int stride, int pad, hipStream_t s) {
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  long long M = (long long)N * H * W;
  const bool narrow = Cin <= 64;
  const int BN_ = narrow ? 64 : 128;
  dim3 grid((Cin + BN_ - 1) / BN_, (M + 127) / 128);
  #define DGRAD(T, BM, BN, WM, WN)                                          \
    hipLaunchKernelGGL((conv_dgrad_kernel<T, BM, BN, WM, WN>), grid,        \
                       dim3(kBlock), 0, s, (const T*)dy, (const T*)w, (T*)dx, sh)
  if (is_bf16) { if (narrow) DGRAD(bf16, 128, 64, 4, 1); else DGRAD(bf16, 128, 128, 2, 2); }
  else { if (narrow) DGRAD(float, 128, 64, 4, 1); else DGRAD(float, 128, 128, 2, 2); }
  #undef DGRAD
}

void launch_conv_wgrad(bool is_bf16, bool out_f32, const void* dy,
                       const void* x, void* dw, int N, int H, int W, int Cin,
                       int Cout, int KH, int KW, int stride, int pad,
                       hipStream_t s) {
  // dw is ALWAYS the fp32 accumulation buffer (bindings allocate zeroed).
  (void)out_f32;
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  int Ntot = KH * KW * Cin;
  const bool narrow_m = Cout <= 64;
  const int BM_ = narrow_m ? 64 : 128;
  int tiles_x = (Ntot + 127) / 128;
  int tiles_y = (Cout + BM_ - 1) / BM_;
  long long Ptot = (long long)N * sh.HO * sh.WO;
  int target = 512 / (tiles_x * tiles_y);
  if (target < 1) target = 1;
  long long max_splits = (Ptot + CBK - 1) / CBK;
  int splits = (int)(max_splits < target ? max_splits : target);
  long long p_chunk = ((Ptot + splits - 1) / splits + CBK - 1) / CBK * CBK;
  splits = (int)((Ptot + p_chunk - 1) / p_chunk);
  dim3 grid(tiles_x, tiles_y, splits);
  #define WGRAD(T, BM, BN, WM, WN)                                          \
    hipLaunchKernelGGL((conv_wgrad_kernel<T, BM, BN, WM, WN>), grid,        \
                       dim3(kBlock), 0, s, (const T*)dy, (const T*)x,       \
                       (float*)dw, sh, p_chunk)
  if (is_bf16) { if (narrow_m) WGRAD(bf16, 64, 128, 1, 4); else WGRAD(bf16, 128, 128, 2, 2); }
  else { if (narrow_m) WGRAD(float, 64, 128, 1, 4); else WGRAD(float, 128, 128, 2, 2); }
  #undef WGRAD
}
