// Implicit-GEMM convolution (gfx950), NHWC layout, forward + dgrad + wgrad.
//
// Required by the ResNet federated configs (SURVEY.md §2.2 row
// "Conv/BatchNorm"; the reference has no conv at all). The convolution IS
// a GEMM on MFMA — no im2col materialization; A/B tiles are gathered
// straight from NHWC tensors into LDS.
//
// Performance structure:
//   * tiles TEMPLATED: 128x128 (2x2 waves), 128x64 (4x1) when the output
//     channel dim is <= 64 (stem/layer1), 64x128 (1x4) for wgrad of
//     64-filter layers;
//   * DOUBLE-BUFFERED LDS, one barrier per k-step;
//   * STATEFUL STAGERS: the pixel decode (m -> n,ho,wo — integer div/mod)
//     is hoisted out of the K-loop (it is k-invariant), and the tap walk
//     (ci,kw,kh / co,kw,kh) advances INCREMENTALLY by CBK per staged tile
//     — the hot loop has no integer division at all on the fast path
//     (channels % 32 == 0, i.e. every ResNet layer but the stem);
//   * wgrad splits the pixel contraction over grid.z (fp32 atomics);
//   * LDS rows padded +8 elements: conflict-free ds_read_b128 groups.
#include "common.h"

constexpr int CBK = 32;
constexpr int CFRAG = 16;

struct ConvShape {
  int N, H, W, Cin, Cout, KH, KW, stride, pad, HO, WO, swz;
};

// ---- stateful stagers ------------------------------------------------------


// bf16 glds staging of a [ROWS][32] K-slice (linear+swizzled image; the
// swizzle rides on each lane's SOURCE address). Only callable when every
// lane's 16 B is in bounds: full row tile, k-slice inside K, rows 16-B
// alignable (ld % 8 == 0).
template <int ROWS>
DEVINL void stage_glds_rows(bf16* __restrict__ lds, const bf16* __restrict__ src,
                            long long ld, long long row0, int k0) {
  const int t = threadIdx.x;
  const int w = t >> 6;
  constexpr int TOTAL = ROWS * 4;          // 16-B slots
#pragma unroll
  for (int p = 0; p * kBlock < TOTAL; ++p) {
    const int idx = p * kBlock + t;
    if (idx >= TOTAL) break;               // whole waves drop out together
    const int row = idx >> 2;
    const int psl = idx & 3;
    const int lsl = psl ^ ((row >> 2) & 3);
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        src + (row0 + row) * ld + k0 + lsl * 8);
    auto l = (__attribute__((address_space(3))) unsigned int*)(
        lds + (long long)(p * kBlock + w * 64) * 8);
    __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  }
}

// Forward A: rows = output pixels, k = (kh,kw,ci), FAST = Cin % 32 == 0.
template <typename T, int ROWS, bool FAST>
struct FwdAStager {
  static constexpr int ELEMS = 16 / (int)sizeof(T);
  static constexpr int TPR = CBK / ELEMS;           // threads per row
  static constexpr int RPP = kBlock / TPR;          // rows per pass
  static constexpr int PASSES = ROWS / RPP;
  int kc;                                           // k offset of this thread
  int row[PASSES];
  int n[PASSES], ho[PASSES], wo[PASSES];
  bool ok[PASSES];
  int ci, kw, kh;                                   // FAST tap state
  int k_generic;                                    // generic-path k cursor
  int m0_;                                          // tile base row
  bool full_;                                       // whole tile in bounds

  DEVINL void init(const ConvShape& sh, int m0, int Mtot) {
    m0_ = m0;
    full_ = (m0 + ROWS <= Mtot);
    kc = (threadIdx.x % TPR) * ELEMS;
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      row[p] = p * RPP + threadIdx.x / TPR;
      const int m = m0 + row[p];
      wo[p] = m % sh.WO;
      const int t = m / sh.WO;
      ho[p] = t % sh.HO;
      n[p] = t / sh.HO;
      ok[p] = m < Mtot;
    }
    ci = kc;     // FAST: Cin >= 32, so k=kc decodes to tap 0, channel kc
    kw = 0;
    kh = 0;
    k_generic = kc;
  }

  // stage one k-tile (called once per tile, ascending), then advance state
  DEVINL void stage(T* __restrict__ lds, const T* __restrict__ x,
                    const ConvShape& sh, int Ktot) {
    using VT = typename VecTraits<T>::VecT;
    if constexpr (sizeof(T) == 2 && FAST) {
      // 1x1 stride-1 conv on a full pixel tile IS a dense GEMM slice
      // (block-uniform condition: the whole 128-pixel tile is in bounds).
      // ci carries the PER-THREAD scalar-path offset kc; the glds helper
      // wants the block-uniform tile base, so strip kc back out.
      if (sh.KH == 1 && sh.stride == 1 && full_ && (sh.Cin % 8) == 0) {
        stage_glds_rows<ROWS>((bf16*)lds, (const bf16*)x, sh.Cin,
                              (long long)m0_, ci - kc);
        ci += CBK;
        if (ci >= sh.Cin) {
          ci -= sh.Cin;
          if (++kw == sh.KW) { kw = 0; ++kh; }
        }
        return;
      }
    }
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      VT v;
      T* vp = reinterpret_cast<T*>(&v);
      if (FAST) {
        const int hi = ho[p] * sh.stride - sh.pad + kh;
        const int wi = wo[p] * sh.stride - sh.pad + kw;
        if (ok[p] && hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W) {
          v = *reinterpret_cast<const VT*>(
              &x[(((long long)n[p] * sh.H + hi) * sh.W + wi) * sh.Cin + ci]);
        } else {
#pragma unroll
          for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
        }
      } else {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j) {
          const int k = k_generic + j;
          vp[j] = (T)0.f;
          if (ok[p] && k < Ktot) {
            const int cij = k % sh.Cin, tap = k / sh.Cin;
            const int kwj = tap % sh.KW, khj = tap / sh.KW;
            const int hi = ho[p] * sh.stride - sh.pad + khj;
            const int wi = wo[p] * sh.stride - sh.pad + kwj;
            if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W)
              vp[j] = x[(((long long)n[p] * sh.H + hi) * sh.W + wi) * sh.Cin + cij];
          }
        }
      }
      *reinterpret_cast<VT*>(&lds[lds_off<T>(row[p], kc)]) = v;
    }
    if (FAST) {
      ci += CBK;
      if (ci >= sh.Cin) {
        ci -= sh.Cin;
        if (++kw == sh.KW) { kw = 0; ++kh; }
      }
    } else {
      k_generic += CBK;
    }
  }
};

// Direct row-major K-contiguous staging (forward B / dgrad B from w_t /
// wgrad A from dy_t): glds for full bf16 tiles, register staging otherwise.
// ld = source row length (may exceed k_limit for split-K pixel slices).
template <typename T, int ROWS>
DEVINL void stage_fwd_B(T* __restrict__ lds, const T* __restrict__ w,
                        int n0, int k0, int Ntot, int k_limit,
                        long long ld) {
  if constexpr (sizeof(T) == 2) {
    if (n0 + ROWS <= Ntot && k0 + CBK <= k_limit && (ld % 8) == 0) {
      stage_glds_rows<ROWS>((bf16*)lds, (const bf16*)w, ld, n0, k0);
      return;
    }
  }
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int TPR = CBK / ELEMS;
  constexpr int RPP = kBlock / TPR;
  using VT = typename VecTraits<T>::VecT;
#pragma unroll
  for (int p = 0; p < ROWS / RPP; ++p) {
    int idx = p * kBlock + threadIdx.x;
    int row = idx / TPR;
    int kc = (idx % TPR) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    if (n0 + row < Ntot && k0 + kc + ELEMS <= k_limit) {
      v = *reinterpret_cast<const VT*>(&w[(long long)(n0 + row) * ld + k0 + kc]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        vp[j] = (n0 + row < Ntot && k0 + kc + j < k_limit)
                    ? w[(long long)(n0 + row) * ld + k0 + kc + j]
                    : (T)0.f;
    }
    *reinterpret_cast<VT*>(&lds[lds_off<T>(row, kc)]) = v;
  }
}

// dgrad A: rows = input pixels, k = (kh,kw,co), FAST = Cout % 32 == 0.
template <typename T, int ROWS, bool FAST>
struct DgradAStager {
  static constexpr int ELEMS = 16 / (int)sizeof(T);
  static constexpr int TPR = CBK / ELEMS;
  static constexpr int RPP = kBlock / TPR;
  static constexpr int PASSES = ROWS / RPP;
  int kc;
  int n[PASSES], hq[PASSES], wq[PASSES];
  bool ok[PASSES];
  int co, kw, kh;
  int k_generic;

  DEVINL void init(const ConvShape& sh, int m0, int Mtot) {
    kc = (threadIdx.x % TPR) * ELEMS;
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int q = m0 + p * RPP + threadIdx.x / TPR;
      wq[p] = q % sh.W;
      const int t = q / sh.W;
      hq[p] = t % sh.H;
      n[p] = t / sh.H;
      ok[p] = q < Mtot;
    }
    co = kc;
    kw = 0;
    kh = 0;
    k_generic = kc;
  }

  DEVINL void stage(T* __restrict__ lds, const T* __restrict__ dy,
                    const ConvShape& sh, int Ktot) {
    using VT = typename VecTraits<T>::VecT;
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      VT v;
      T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
      for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
      if (FAST) {
        if (ok[p]) {
          const int hnum = hq[p] + sh.pad - kh, wnum = wq[p] + sh.pad - kw;
          if (hnum >= 0 && wnum >= 0 && hnum % sh.stride == 0 &&
              wnum % sh.stride == 0) {
            const int ho = hnum / sh.stride, wo = wnum / sh.stride;
            if (ho < sh.HO && wo < sh.WO)
              v = *reinterpret_cast<const VT*>(
                  &dy[(((long long)n[p] * sh.HO + ho) * sh.WO + wo) * sh.Cout + co]);
          }
        }
      } else if (ok[p]) {
#pragma unroll
        for (int j = 0; j < ELEMS; ++j) {
          const int k = k_generic + j;
          if (k < Ktot) {
            const int coj = k % sh.Cout, tap = k / sh.Cout;
            const int kwj = tap % sh.KW, khj = tap / sh.KW;
            const int hnum = hq[p] + sh.pad - khj, wnum = wq[p] + sh.pad - kwj;
            if (hnum >= 0 && wnum >= 0 && hnum % sh.stride == 0 &&
                wnum % sh.stride == 0) {
              const int ho = hnum / sh.stride, wo = wnum / sh.stride;
              if (ho < sh.HO && wo < sh.WO)
                vp[j] = dy[(((long long)n[p] * sh.HO + ho) * sh.WO + wo) * sh.Cout + coj];
            }
          }
        }
      }
      *reinterpret_cast<VT*>(&lds[lds_off<T>(p * RPP + threadIdx.x / TPR, kc)]) = v;
    }
    if (FAST) {
      co += CBK;
      if (co >= sh.Cout) {
        co -= sh.Cout;
        if (++kw == sh.KW) { kw = 0; ++kh; }
      }
    } else {
      k_generic += CBK;
    }
  }
};

// wgrad B: rows = (kh,kw,ci) — tap decode HOISTED (k-invariant); the pixel
// (= contraction index) advances incrementally across staged tiles.
template <typename T, int ROWS>
struct WgradBStager {
  static constexpr int ELEMS = 16 / (int)sizeof(T);
  static constexpr int VPK = ROWS / ELEMS;
  static constexpr int TOTAL = CBK * VPK;
  static constexpr int PASSES = (TOTAL + kBlock - 1) / kBlock;
  int k_in_tile[PASSES], rr[PASSES], r_local[PASSES];
  int ci[PASSES], kw[PASSES], kh[PASSES];
  bool active[PASSES], one_tap[PASSES], r_ok[PASSES];
  // pixel state shared across passes with differing k offsets: track per pass
  int wo[PASSES], ho[PASSES], nn[PASSES];
  bool p_ok[PASSES];

  DEVINL void init(const ConvShape& sh, int n0, long long p0, long long p_limit,
                   int Rtot) {
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int idx = p * kBlock + threadIdx.x;
      active[p] = idx < TOTAL;
      // r-run fastest: consecutive lanes read consecutive 16-B ci runs of
      // the SAME pixel (coalesced gather); pixel-fastest order scattered
      // adjacent lanes a whole pixel stride apart
      k_in_tile[p] = idx / VPK;
      r_local[p] = (idx % VPK) * ELEMS;
      rr[p] = n0 + r_local[p];
      r_ok[p] = rr[p] < Rtot;
      ci[p] = r_ok[p] ? rr[p] % sh.Cin : 0;
      const int tap = r_ok[p] ? rr[p] / sh.Cin : 0;
      kw[p] = tap % sh.KW;
      kh[p] = tap / sh.KW;
      one_tap[p] = (rr[p] / sh.Cin) == ((rr[p] + ELEMS - 1) / sh.Cin);
      const long long px = p0 + k_in_tile[p];
      p_ok[p] = px < p_limit;
      wo[p] = (int)(px % sh.WO);
      const long long t = px / sh.WO;
      ho[p] = (int)(t % sh.HO);
      nn[p] = (int)(t / sh.HO);
    }
  }

  DEVINL void stage(T* __restrict__ lds, const T* __restrict__ x,
                    const ConvShape& sh) {
    using VT = typename VecTraits<T>::VecT;
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      if (!active[p]) continue;
      VT v;
      T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
      for (int j = 0; j < ELEMS; ++j) vp[j] = (T)0.f;
      if (p_ok[p] && r_ok[p]) {
        if (one_tap[p]) {
          const int hi = ho[p] * sh.stride - sh.pad + kh[p];
          const int wi = wo[p] * sh.stride - sh.pad + kw[p];
          if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W)
            v = *reinterpret_cast<const VT*>(
                &x[(((long long)nn[p] * sh.H + hi) * sh.W + wi) * sh.Cin + ci[p]]);
        } else {
#pragma unroll
          for (int j = 0; j < ELEMS; ++j) {
            const int rj = rr[p] + j;
            const int cij = rj % sh.Cin, tapj = rj / sh.Cin;
            const int kwj = tapj % sh.KW, khj = tapj / sh.KW;
            const int hi = ho[p] * sh.stride - sh.pad + khj;
            const int wi = wo[p] * sh.stride - sh.pad + kwj;
            if (hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W)
              vp[j] = x[(((long long)nn[p] * sh.H + hi) * sh.W + wi) * sh.Cin + cij];
          }
        }
      }
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        lds[lds_off<T>(r_local[p] + j, k_in_tile[p])] = vp[j];
      // advance pixel by CBK
      wo[p] += CBK;
      while (wo[p] >= sh.WO) {
        wo[p] -= sh.WO;
        if (++ho[p] == sh.HO) { ho[p] = 0; ++nn[p]; }
      }
    }
  }

  DEVINL void set_p_ok(long long p0, long long p_limit) {
#pragma unroll
    for (int p = 0; p < PASSES; ++p) p_ok[p] = (p0 + k_in_tile[p]) < p_limit;
  }
};

// ---- MFMA compute ----------------------------------------------------------

template <typename T, int MF, int NF>
DEVINL void conv_mma(const T* a_lds, const T* b_lds, f32x4 (&acc)[MF][NF],
                     int lane, int wm0, int wn0) {
  if constexpr (sizeof(T) == 2) {
    s16x8 a_frag[MF], b_frag[NF];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &a_lds[lds_off<T>(wm0 + mf * CFRAG + (lane & 15), (lane >> 4) * 8)]);
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &b_lds[lds_off<T>(wn0 + nf * CFRAG + (lane & 15), (lane >> 4) * 8)]);
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
        a_s[mf] = ((const float*)a_lds)[lds_off<float>(wm0 + mf * CFRAG + (lane & 15), kidx)];
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        b_s[nf] = ((const float*)b_lds)[lds_off<float>(wn0 + nf * CFRAG + (lane & 15), kidx)];
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

template <typename T, int BM, int BN, int WAVES_M, int WAVES_N, bool FAST>
__global__ __launch_bounds__(kBlock) void conv_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ y,
    ConvShape sh) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * lds_row_elems<T>()];
  __shared__ T b_lds[2][BN * lds_row_elems<T>()];
  const int Mtot = sh.N * sh.HO * sh.WO;
  const int Ntot = sh.Cout;
  const int Ktot = sh.KH * sh.KW * sh.Cin;
  // XCD-aware remap (see gemm.hip)
  int t_n, t_m;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (sh.swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    t_n = bid % gridDim.x;
    t_m = bid / gridDim.x;
  }
  const int m0 = t_m * BM, n0 = t_n * BN;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM, wn0 = (wid % WAVES_N) * WN;
  f32x4 acc[MF][NF] = {};
  const int nk = (Ktot + CBK - 1) / CBK;
  FwdAStager<T, BM, FAST> sa;
  sa.init(sh, m0, Mtot);
  sa.stage(a_lds[0], x, sh, Ktot);
  stage_fwd_B<T, BN>(b_lds[0], w, n0, 0, Ntot, Ktot, Ktot);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      sa.stage(a_lds[cur ^ 1], x, sh, Ktot);
      stage_fwd_B<T, BN>(b_lds[cur ^ 1], w, n0, (kt + 1) * CBK, Ntot, Ktot, Ktot);
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

template <typename T, int BM, int BN, int WAVES_M, int WAVES_N, bool FAST>
__global__ __launch_bounds__(kBlock) void conv_dgrad_kernel(
    const T* __restrict__ dy, const T* __restrict__ w_t, T* __restrict__ dx,
    ConvShape sh) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * lds_row_elems<T>()];
  __shared__ T b_lds[2][BN * lds_row_elems<T>()];
  const int Mtot = sh.N * sh.H * sh.W;
  const int Ntot = sh.Cin;
  const int Ktot = sh.KH * sh.KW * sh.Cout;
  // XCD-aware remap (see gemm.hip)
  int t_n, t_m;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (sh.swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    t_n = bid % gridDim.x;
    t_m = bid / gridDim.x;
  }
  const int m0 = t_m * BM, n0 = t_n * BN;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM, wn0 = (wid % WAVES_N) * WN;
  f32x4 acc[MF][NF] = {};
  const int nk = (Ktot + CBK - 1) / CBK;
  DgradAStager<T, BM, FAST> sa;
  sa.init(sh, m0, Mtot);
  sa.stage(a_lds[0], dy, sh, Ktot);
  // B from the TRANSPOSED weight copy w_t [Cin][KH*KW*Cout] (row-major,
  // k-contiguous: same staging as forward B, glds-eligible)
  stage_fwd_B<T, BN>(b_lds[0], w_t, n0, 0, Ntot, Ktot, Ktot);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      sa.stage(a_lds[cur ^ 1], dy, sh, Ktot);
      stage_fwd_B<T, BN>(b_lds[cur ^ 1], w_t, n0, (kt + 1) * CBK, Ntot, Ktot, Ktot);
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

// Split-K over pixels (see launcher): partials -> fp32 atomics.
template <typename T, int BM, int BN, int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(kBlock) void conv_wgrad_kernel(
    const T* __restrict__ dy_t, const T* __restrict__ x, float* __restrict__ dw,
    ConvShape sh, long long p_chunk) {
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;
  constexpr int MF = WM / CFRAG, NF = WN / CFRAG;
  __shared__ T a_lds[2][BM * lds_row_elems<T>()];
  __shared__ T b_lds[2][BN * lds_row_elems<T>()];
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
  WgradBStager<T, BN> sb;
  sb.init(sh, n0, p_begin, p_end, Ntot);
  // A from the TRANSPOSED dy copy [Cout][P] (k = pixel contiguous): plain
  // direct staging (glds-eligible) instead of scalar transposed writes
  stage_fwd_B<T, BM>(a_lds[0], dy_t, m0, (int)p_begin, Mtot, (int)p_end, Ptot);
  sb.stage(b_lds[0], x, sh);
  __syncthreads();
  for (long long kt = 0; kt < nk; ++kt) {
    const int cur = (int)(kt & 1);
    if (kt + 1 < nk) {
      stage_fwd_B<T, BM>(a_lds[cur ^ 1], dy_t, m0,
                         (int)(p_begin + (kt + 1) * CBK), Mtot, (int)p_end,
                         Ptot);
      sb.set_p_ok(p_begin + (kt + 1) * CBK, p_end);
      sb.stage(b_lds[cur ^ 1], x, sh);
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

// ---- tap-replicated wgrad (3x3 stride-1 pad-1, the ResNet hot case) -------
//
// The generic wgrad above is TRAFFIC-bound: its B operand re-gathers x once
// per (kh,kw) output column tile (~9x HBM traffic) and re-reads dy_t once
// per N-tile. This kernel stages one 32-pixel dy_t tile and ONE x read per
// (pixel, ci) from global, writes it into NINE per-tap LDS images (the
// overlapping gathers dedup in L2), and contracts all 9 taps against the
// same dy tile — x and dy each cross HBM ~once per (co,ci) tile combo
// instead of ~9x/~5x. Per-tap images keep every fragment read a b128 on
// the standard swizzled image (a single halo'd window would put tap
// windows at 2-B shifts — unaligned for ds_read_b128 and for
// ds_read_b64_tr_b16 alike).
//
// Block: 512 threads (8 waves as 2(co) x 4(ci)), computes
// dw[64 co][9 taps][64 ci] (72 acc VGPRs/lane); K = pixels, 32 per step,
// split over grid.z into fp32 atomics. Eligibility (launcher): bf16,
// 3x3/stride1/pad1, Cin%64==0, Cout%64==0, Ptot%32==0, 32%W==0 (flat
// pixel advance keeps wo per-thread constant).
namespace wt9 {

constexpr int PXK = 32;                 // pixels per k-step
constexpr int NTAP = 9;
constexpr int THREADS = 512;
constexpr int IMG = 64 * PXK;           // one [64][32] swizzled image

// x-image offset: COMBINED slot XOR (row>>2 ^ row>>3). The b128 fragment
// read shares banks among rows {r, r+4, r+8, r+12}, whose combined bits
// are all distinct (conflict-free read, same as the plain row>>2 form);
// the transposed flush writes rows 8j+q with q fixed per instruction, so
// only row>>3 (= j) varies per lane — the combined XOR spreads the write
// group over 4 slots where row>>2 alone reaches 2 (measured 4-way).
DEVINL int xoff(int row, int col) {
  const int sl = col >> 3;
  return row * PXK + (((sl ^ (row >> 2) ^ (row >> 3)) & 3) << 3) + (col & 7);
}

// x gather state: 3 passes cover 9 taps x 16 px-PAIRS x 8 ci-runs = 1152
// slots; each slot loads TWO adjacent pixels so the transposed flush can
// write (px, px+1) element pairs as single b32s — half the ds_write
// instructions of the scalar-b16 form (the write pattern is a transpose,
// so per-instruction lanes share a row parity and collapse onto 16 of the
// 32 banks; pairing is the lever that halves the conflict-serialized
// instruction count without breaking the b128 read image).
struct XStager {
  static constexpr int SLOTS = NTAP * (PXK / 4) * 8;   // px QUADS
  static constexpr int PASSES = (SLOTS + THREADS - 1) / THREADS;
  // only the pixel-walk state (for the BASE pixel of the quad) lives in
  // registers; tap/quad/ci-run are recomputed from the slot index. The
  // quad never crosses a row: the launcher gates W % 4 == 0 and quads are
  // 4-aligned, so all 4 pixels share (n, ho).
  int n[PASSES], ho[PASSES], wo[PASSES];

  DEVINL void init(const ConvShape& sh, long long p0) {
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int idx = p * THREADS + threadIdx.x;
      const int px = ((idx & 63) >> 3) * 4;
      const long long g = p0 + px;
      wo[p] = (int)(g % sh.W);          // WO == W (stride 1, pad 1, 3x3)
      const long long t = g / sh.W;
      ho[p] = (int)(t % sh.H);
      n[p] = (int)(t / sh.H);
    }
  }

  // T14 register staging (guide: register-staged operands when the LDS
  // write pattern is scattered): load() ISSUES the global gathers for one
  // k-step into registers and advances the pixel walk; flush() writes the
  // previously loaded registers to LDS one full k-step later, so the
  // global latency hides behind a whole step of MFMA instead of stalling
  // the load->ds_write chain. Four pixels per slot: the transposed flush
  // packs them as ONE ds_write_b64 — half the b32-pair form's write
  // instructions, and with the combined slot XOR (see xoff) the write
  // group lands on ~2x the banks (the b32-pair flush measured 4-way
  // conflict-bound; row parity is per-instruction constant, so only the
  // row>>2 / row>>3 bits can spread it).
  s16x8 v0[PASSES], v1[PASSES], v2[PASSES], v3[PASSES];

  DEVINL void load(const bf16* __restrict__ x, const ConvShape& sh, int ci0) {
    const int hstep = PXK / sh.W;       // 32 % W == 0 (gate)
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int idx = p * THREADS + threadIdx.x;
      if (idx >= SLOTS) continue;
      const int tap = idx >> 6;
      const int j = idx & 7;
      const int dh = tap / 3, dw = tap % 3;
      const int hi = ho[p] + dh - 1;
      const bool hok = hi >= 0 && hi < sh.H;
      const long long rowbase =
          (((long long)n[p] * sh.H + hi) * sh.W) * sh.Cin + ci0 + j * 8;
      s16x8* vs[4] = {&v0[p], &v1[p], &v2[p], &v3[p]};
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        const int wi = wo[p] + d + dw - 1;
        if (hok && wi >= 0 && wi < sh.W) {
          *vs[d] = *reinterpret_cast<const s16x8*>(
              &x[rowbase + (long long)wi * sh.Cin]);
        } else {
          *vs[d] = s16x8{};
        }
      }
      ho[p] += hstep;
      while (ho[p] >= sh.H) { ho[p] -= sh.H; ++n[p]; }
    }
  }

  DEVINL void flush(bf16* __restrict__ xlds) {
#pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int idx = p * THREADS + threadIdx.x;
      if (idx >= SLOTS) continue;
      const int tap = idx >> 6;
      const int px = ((idx & 63) >> 3) * 4;
      const int j = idx & 7;
      bf16* img = xlds + tap * IMG;
      const bf16* p0 = reinterpret_cast<const bf16*>(&v0[p]);
      const bf16* p1 = reinterpret_cast<const bf16*>(&v1[p]);
      const bf16* p2 = reinterpret_cast<const bf16*>(&v2[p]);
      const bf16* p3 = reinterpret_cast<const bf16*>(&v3[p]);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const unsigned w0 = (unsigned short)__bfloat16_as_ushort(p0[q]) |
                            ((unsigned)(unsigned short)__bfloat16_as_ushort(p1[q]) << 16);
        const unsigned w1 = (unsigned short)__bfloat16_as_ushort(p2[q]) |
                            ((unsigned)(unsigned short)__bfloat16_as_ushort(p3[q]) << 16);
        *reinterpret_cast<uint2*>(&img[xoff(j * 8 + q, px)]) = uint2{w0, w1};
      }
    }
  }
};

// dy_t [Cout][P] tile [64 rows][32 px] via glds (rows 16-B aligned:
// P % 32 == 0); swizzle rides on the source slot address
DEVINL void stage_dy(bf16* __restrict__ lds, const bf16* __restrict__ dy_t,
                     long long P, int co0, long long p0) {
  const int t = threadIdx.x;
  if (t < 256) {
    const int row = t >> 2, psl = t & 3;
    const int lsl = psl ^ ((row >> 2) & 3);
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        dy_t + (long long)(co0 + row) * P + p0 + lsl * 8);
    auto l = (__attribute__((address_space(3))) unsigned int*)(
        lds + (long long)t * 8);
    __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  }
}

__global__ __launch_bounds__(THREADS) void wgrad_tap_kernel(
    const bf16* __restrict__ dy_t, const bf16* __restrict__ x,
    float* __restrict__ dw, ConvShape sh, long long p_chunk) {
  // ONE shared object: [2 buffers][dy image + 9 x images]
  __shared__ bf16 lds_all[2 * (1 + NTAP) * IMG];
  auto dybuf = [&](int b) -> bf16* { return lds_all + b * (1 + NTAP) * IMG; };
  auto xbuf = [&](int b) -> bf16* {
    return lds_all + b * (1 + NTAP) * IMG + IMG;
  };

  const long long Ptot = (long long)sh.N * sh.H * sh.W;
  const long long p_begin = (long long)blockIdx.z * p_chunk;
  const long long p_end = min(p_begin + p_chunk, Ptot);
  const long long nk = (p_end - p_begin) / PXK;
  if (nk <= 0) return;
  const int ci0 = blockIdx.x * 64, co0 = blockIdx.y * 64;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm0 = (wid >> 2) * 32;      // co half
  const int wn0 = (wid & 3) * 16;       // ci quarter

  f32x4 acc[2][NTAP] = {};

  XStager sx;
  sx.init(sh, p_begin);
  sx.load(x, sh, ci0);                  // gathers for k-step 0
  sx.flush(xbuf(0));
  stage_dy(dybuf(0), dy_t, Ptot, co0, p_begin);
  if (nk > 1) sx.load(x, sh, ci0);      // k-step 1 in flight
  // counted wait + RAW barrier: __syncthreads() would vmcnt(0)-drain the
  // register loads just issued; allow EXACTLY the in-flight loads to stay
  // outstanding and drain the dy glds (a count larger than what is
  // actually in flight would skip the glds drain — vmcnt(N) is a no-op
  // when fewer than N+1 ops are pending)
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  if (nk > 1) {
    // wave 0 owns the 2nd staging pass (8 gathers in flight), others 4;
    // the count still drains the (older) dy glds
    if (threadIdx.x < 64)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  for (long long kt = 0; kt < nk; ++kt) {
    const int cur = (int)(kt & 1);
    if (kt + 1 < nk) {
      stage_dy(dybuf(cur ^ 1), dy_t, Ptot, co0, p_begin + (kt + 1) * PXK);
      sx.flush(xbuf(cur ^ 1));          // registers loaded one step ago
      if (kt + 2 < nk) sx.load(x, sh, ci0);
    }
    const bf16* dyl = dybuf(cur);
    const bf16* xl = xbuf(cur);
    s16x8 a_frag[2], b_frag[NTAP];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &dyl[lds_off<bf16>(wm0 + mf * 16 + (lane & 15), (lane >> 4) * 8)]);
#pragma unroll
    for (int t = 0; t < NTAP; ++t)
      b_frag[t] = *reinterpret_cast<const s16x8*>(
          &xl[t * IMG + xoff(wn0 + (lane & 15), (lane >> 4) * 8)]);
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int t = 0; t < NTAP; ++t)
        acc[mf][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[t], acc[mf][t], 0, 0, 0);
    // end-of-step barrier: next buffer's dy glds has had the whole compute
    // phase to land; spare the 6 k+2 register gathers ONLY when they were
    // issued, else drain fully (see prologue comment); flush's ds_writes
    // ordered by lgkmcnt(0)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (kt + 2 < nk) {
      if (threadIdx.x < 64)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }
  // drain every outstanding VMEM (glds has no register dep the compiler
  // would wait on) before the epilogue
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int col_in_frag = lane & 15, row_base = (lane >> 4) * 4;
  const long long ldw = (long long)NTAP * sh.Cin;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int t = 0; t < NTAP; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = co0 + wm0 + mf * 16 + row_base + r;
        const long long col = (long long)t * sh.Cin + ci0 + wn0 + col_in_frag;
        if (gridDim.z == 1)
          dw[row * ldw + col] = acc[mf][t][r];
        else
          atomicAdd(&dw[row * ldw + col], acc[mf][t][r]);
      }
}

}  // namespace wt9

// Instantiations: Big = 128x128 (2x2), NarrowN = 128x64 (4x1),
// NarrowM (wgrad) = 64x128 (1x4); FAST per channel divisibility.
#define INST_CONV(T)                                                        \
  template __global__ void conv_fwd_kernel<T, 128, 128, 2, 2, true>(        \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_fwd_kernel<T, 128, 128, 2, 2, false>(       \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_fwd_kernel<T, 128, 64, 4, 1, true>(         \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_fwd_kernel<T, 128, 64, 4, 1, false>(        \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 128, 2, 2, true>(      \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 128, 2, 2, false>(     \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 64, 4, 1, true>(       \
      const T*, const T*, T*, ConvShape);                                   \
  template __global__ void conv_dgrad_kernel<T, 128, 64, 4, 1, false>(      \
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
  // XCD swizzle only past the 256 MiB L3 (see gemm.hip)
  long long act = (long long)N * H * W * Cin + (long long)N * sh.HO * sh.WO * Cout;
  sh.swz = act * 2 > (200LL << 20) ? 1 : 0;
  return sh;
}

void launch_conv_fwd(bool is_bf16, const void* x, const void* w, void* y,
                     int N, int H, int W, int Cin, int Cout, int KH, int KW,
                     int stride, int pad, hipStream_t s) {
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  long long M = (long long)N * sh.HO * sh.WO;
  const bool narrow = Cout <= 64;
  const bool fast = (Cin % CBK) == 0;
  const int BN_ = narrow ? 64 : 128;
  dim3 grid((Cout + BN_ - 1) / BN_, (M + 127) / 128);
  #define FWD(T, BM, BN, WM, WN, F)                                         \
    hipLaunchKernelGGL((conv_fwd_kernel<T, BM, BN, WM, WN, F>), grid,       \
                       dim3(kBlock), 0, s, (const T*)x, (const T*)w, (T*)y, sh)
  if (is_bf16) {
    if (narrow) { if (fast) FWD(bf16, 128, 64, 4, 1, true); else FWD(bf16, 128, 64, 4, 1, false); }
    else { if (fast) FWD(bf16, 128, 128, 2, 2, true); else FWD(bf16, 128, 128, 2, 2, false); }
  } else {
    if (narrow) { if (fast) FWD(float, 128, 64, 4, 1, true); else FWD(float, 128, 64, 4, 1, false); }
    else { if (fast) FWD(float, 128, 128, 2, 2, true); else FWD(float, 128, 128, 2, 2, false); }
  }
  #undef FWD
}

void launch_conv_dgrad(bool is_bf16, const void* dy, const void* w, void* dx,
                       int N, int H, int W, int Cin, int Cout, int KH, int KW,
                       int stride, int pad, hipStream_t s) {
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  long long M = (long long)N * H * W;
  const bool narrow = Cin <= 64;
  const bool fast = (Cout % CBK) == 0;
  const int BN_ = narrow ? 64 : 128;
  dim3 grid((Cin + BN_ - 1) / BN_, (M + 127) / 128);
  #define DG(T, BM, BN, WM, WN, F)                                          \
    hipLaunchKernelGGL((conv_dgrad_kernel<T, BM, BN, WM, WN, F>), grid,     \
                       dim3(kBlock), 0, s, (const T*)dy, (const T*)w, (T*)dx, sh)
  if (is_bf16) {
    if (narrow) { if (fast) DG(bf16, 128, 64, 4, 1, true); else DG(bf16, 128, 64, 4, 1, false); }
    else { if (fast) DG(bf16, 128, 128, 2, 2, true); else DG(bf16, 128, 128, 2, 2, false); }
  } else {
    if (narrow) { if (fast) DG(float, 128, 64, 4, 1, true); else DG(float, 128, 64, 4, 1, false); }
    else { if (fast) DG(float, 128, 128, 2, 2, true); else DG(float, 128, 128, 2, 2, false); }
  }
  #undef DG
}

void launch_conv_wgrad(bool is_bf16, bool out_f32, const void* dy,
                       const void* x, void* dw, int N, int H, int W, int Cin,
                       int Cout, int KH, int KW, int stride, int pad,
                       hipStream_t s) {
  // dw is ALWAYS the fp32 accumulation buffer (bindings allocate zeroed).
  (void)out_f32;
  ConvShape sh = make_shape(N, H, W, Cin, Cout, KH, KW, stride, pad);
  // tap-replicated kernel for the ResNet hot case (one x/dy HBM pass per
  // (co,ci) tile combo instead of per output column tile — see wt9)
  if (is_bf16 && KH == 3 && KW == 3 && stride == 1 && pad == 1 &&
      (Cin % 64) == 0 && (Cout % 64) == 0) {
    long long Ptot = (long long)N * sh.HO * sh.WO;
    if (Ptot % wt9::PXK == 0 && (wt9::PXK % W) == 0 && (W % 4) == 0) {
      int gx = Cin / 64, gy = Cout / 64;
      int target = 1024 / (gx * gy);
      if (target < 1) target = 1;
      long long maxs = Ptot / wt9::PXK;
      int splits = (int)(maxs < target ? maxs : target);
      long long p_chunk =
          ((Ptot / wt9::PXK + splits - 1) / splits) * wt9::PXK;
      splits = (int)((Ptot + p_chunk - 1) / p_chunk);
      dim3 grid(gx, gy, splits);
      hipLaunchKernelGGL(wt9::wgrad_tap_kernel, grid, dim3(wt9::THREADS), 0,
                         s, (const bf16*)dy, (const bf16*)x, (float*)dw, sh,
                         p_chunk);
      return;
    }
  }
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
  #define WG(T, BM, BN, WM, WN)                                             \
    hipLaunchKernelGGL((conv_wgrad_kernel<T, BM, BN, WM, WN>), grid,        \
                       dim3(kBlock), 0, s, (const T*)dy, (const T*)x,       \
                       (float*)dw, sh, p_chunk)
  if (is_bf16) { if (narrow_m) WG(bf16, 64, 128, 1, 4); else WG(bf16, 128, 128, 2, 2); }
  else { if (narrow_m) WG(float, 64, 128, 1, 4); else WG(float, 128, 128, 2, 2); }
  #undef WG
}
