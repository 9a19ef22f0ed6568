// MFMA GEMM kernels (gfx950) — the Linear-layer hot path and the base of
// the attention path.
//
// From-scratch CDNA4 design (the reference's only GEMM is CPU nn.Linear,
// /root/reference/demo.py:23):
//   * v_mfma_f32_16x16x32_bf16 (bf16 in, fp32 acc) / v_mfma_f32_16x16x4_f32
//     (exact f32 — no TF32 on gfx950);
//   * tile geometry templated: 128x128 (2x2 waves), 128x64 (4x1), 64x128
//     (1x4) — the launcher picks whichever fills 256 CUs;
//   * DOUBLE-BUFFERED LDS, one barrier per K-step (2-phase pipeline);
//   * bf16 staging uses __builtin_amdgcn_global_load_lds (16-B direct
//     HBM->LDS DMA, no VGPR round-trip) on full interior tiles. glds
//     writes lane-linear, so the LDS image is LINEAR (no pad) and bank
//     conflicts are killed by a slot XOR swizzle applied on BOTH the
//     per-lane global SOURCE address and the ds_read offsets (guide rule
//     21: dest stays linear). Layout: 64-B rows (BK=32 bf16), 16-B slot
//     s at row r lives at physical slot s ^ ((r>>2)&3) — the 16-lane
//     ds_read_b128 fragment groups then touch 16 distinct slots per 256-B
//     bank row (conflict-free);
//   * edge tiles / transposed operands fall back to register staging that
//     writes the same swizzled image; fp32 uses a padded layout (+2
//     elements) with scalar b32 fragment reads;
//   * batched via grid.z (attention: one batch per B*H);
//   * TN split-K variant for Linear wgrad (small output, long contraction).
#include "common.h"
#include <cstdlib>

constexpr int BK = 32;
constexpr int FRAG = 16;
constexpr int BM = 128, BN = 128;   // default tile (launcher may narrow)

// LDS layout helpers (lds_off / lds_row_elems) live in common.h.

// ---- staging ---------------------------------------------------------------
// Canonical image: rows = output-dim (M or N), cols = K slice (k-contig).

// direct: source row-major [rows][K] -> vector loads; writes swizzled image.
template <typename T, int ROWS>
DEVINL void stage_direct(T* __restrict__ lds, const T* __restrict__ src,
                         long long ld, int row0, int k0, int rows_limit,
                         int k_limit) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int THREADS_PER_ROW = BK / ELEMS;
  constexpr int TOTAL = ROWS * THREADS_PER_ROW;       // vector slots in tile
  using VT = typename VecTraits<T>::VecT;
#pragma unroll
  for (int p = 0; p < (TOTAL + kBlock - 1) / kBlock; ++p) {
    int idx = p * kBlock + threadIdx.x;
    if (idx >= TOTAL) break;
    int row = idx / THREADS_PER_ROW;
    int kc = (idx % THREADS_PER_ROW) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    if (row0 + row < rows_limit && k0 + kc + ELEMS <= k_limit) {
      v = *reinterpret_cast<const VT*>(&src[(long long)(row0 + row) * ld + k0 + kc]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        vp[j] = (row0 + row < rows_limit && k0 + kc + j < k_limit)
                    ? src[(long long)(row0 + row) * ld + k0 + kc + j]
                    : (T)0.f;
    }
    *reinterpret_cast<VT*>(&lds[lds_off<T>(row, kc)]) = v;
  }
}

// glds: bf16 full tiles with 16-B-alignable rows. The hardware writes lane
// l's 16 B at (wave-uniform base) + l*16, so the swizzle is applied to the
// SOURCE address each lane loads from.
DEVINL void stage_direct_glds(bf16* __restrict__ lds, const bf16* __restrict__ src,
                              long long ld, int row0, int k0, int rows) {
  const int t = threadIdx.x;
  const int w = t >> 6;                  // wave id (uniform per wave)
  const int total = rows * 4;            // 16-B slots in the tile
  for (int p = 0; p * kBlock < total; ++p) {
    const int idx = p * kBlock + t;
    if (idx >= total) break;             // whole waves drop out together
    const int row = idx >> 2;                // 4 slots per row
    const int psl = idx & 3;                 // physical slot this lane fills
    const int lsl = psl ^ ((row >> 2) & 3);  // logical slot -> source k
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        src + (long long)(row0 + row) * ld + k0 + lsl * 8);
    // wave-uniform LDS base: this wave's 64 lanes fill 1 KiB linearly
    auto l = (__attribute__((address_space(3))) unsigned int*)(
        lds + (long long)(p * kBlock + w * 64) * 8);
    __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  }
}

// transposed: source row-major [K][rows] -> vector load along rows, scatter
// scalar writes into the (swizzled/padded) image.
template <typename T, int ROWS>
DEVINL void stage_transposed(T* __restrict__ lds, const T* __restrict__ src,
                             long long ld, int row0, int k0, int rows_limit,
                             int k_limit) {
  constexpr int ELEMS = 16 / sizeof(T);
  constexpr int VECS_PER_K = ROWS / ELEMS;
  using VT = typename VecTraits<T>::VecT;
  constexpr int TOTAL = BK * VECS_PER_K;
#pragma unroll
  for (int p = 0; p < (TOTAL + kBlock - 1) / kBlock; ++p) {
    int idx = p * kBlock + threadIdx.x;
    if (idx >= TOTAL) break;
    int k = idx % BK;
    int r = (idx / BK) * ELEMS;
    VT v;
    T* vp = reinterpret_cast<T*>(&v);
    if (k0 + k < k_limit && row0 + r + ELEMS <= rows_limit) {
      v = *reinterpret_cast<const VT*>(&src[(long long)(k0 + k) * ld + row0 + r]);
    } else {
#pragma unroll
      for (int j = 0; j < ELEMS; ++j)
        vp[j] = (k0 + k < k_limit && row0 + r + j < rows_limit)
                    ? src[(long long)(k0 + k) * ld + row0 + r + j]
                    : (T)0.f;
    }
#pragma unroll
    for (int j = 0; j < ELEMS; ++j) lds[lds_off<T>(r + j, k)] = vp[j];
  }
}

// ---- MFMA over one staged K-step -------------------------------------------

template <typename T, int MF, int NF>
DEVINL void gemm_mma(const T* a_lds, const T* b_lds, f32x4 (&acc)[MF][NF],
                     int lane, int wm0, int wn0) {
  if constexpr (sizeof(T) == 2) {
    s16x8 a_frag[MF], b_frag[NF];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &a_lds[lds_off<T>(wm0 + mf * FRAG + (lane & 15), (lane >> 4) * 8)]);
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &b_lds[lds_off<T>(wn0 + nf * FRAG + (lane & 15), (lane >> 4) * 8)]);
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
  } else {
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      float a_s[MF], b_s[NF];
      const int kidx = kk * 4 + (lane >> 4);
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
        a_s[mf] = ((const float*)a_lds)[lds_off<T>(wm0 + mf * FRAG + (lane & 15), kidx)];
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        b_s[nf] = ((const float*)b_lds)[lds_off<T>(wn0 + nf * FRAG + (lane & 15), kidx)];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_s[mf], b_s[nf], acc[mf][nf], 0, 0, 0);
    }
  }
}

// ---- the kernel ------------------------------------------------------------
// C[M,N] = alpha * op(A) @ op(B) + beta * C (+ bias[n]) (+ relu)
//   TA=0: A[M,K] row-major; TA=1: A[K,M] row-major.
//   TB=0: B[K,N] row-major; TB=1: B[N,K] row-major.

template <typename T, typename TOUT, bool TA, bool TB, bool RELU,
          int BM_ = BM, int BN_ = BN, int WAVES_M = 2, int WAVES_N = 2,
          bool SPLITK = false>
__global__ __launch_bounds__(kBlock) void gemm_kernel(
    const T* __restrict__ A, const T* __restrict__ B, TOUT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, float alpha,
    float beta, long long strideA, long long strideB, long long strideC,
    int k_chunk = 0, int use_swz = 0, int b_group = 1,
    GemmStrides gs = GemmStrides{0, 0, 0, 1, 0, 0, 0}) {
  constexpr int WM = BM_ / WAVES_M;
  constexpr int WN = BN_ / WAVES_N;
  constexpr int MF = WM / FRAG;
  constexpr int NF = WN / FRAG;
  constexpr int RE = lds_row_elems<T>();
  // ONE shared object (glds pipelines de-schedule with several — guide
  // §5 trap (a)); [buffer][A image | B image].
  __shared__ T lds_all[2 * (BM_ + BN_) * RE];
  auto a_lds = [&](int buf) -> T* { return lds_all + buf * (BM_ + BN_) * RE; };
  auto b_lds = [&](int buf) -> T* {
    return lds_all + buf * (BM_ + BN_) * RE + BM_ * RE;
  };

  if (gs.heads > 1) {
    // two-level batch: z = outer * heads + head (strided attention views)
    const long long zo = blockIdx.z / gs.heads;
    const long long zi = blockIdx.z % gs.heads;
    A += zo * strideA + zi * gs.a2;
    B += zo * strideB + (zi / b_group) * gs.b2;   // GQA on the head index
    C += zo * strideC + zi * gs.c2;
  } else {
    A += (long long)blockIdx.z * strideA;
    // b_group > 1: GQA — b_group consecutive batches share one B (KV head)
    B += (long long)(blockIdx.z / b_group) * strideB;
    C += (long long)blockIdx.z * strideC;
  }
  // XCD-aware remap (T1): the dispatcher places block b on XCD b%8, so
  // consecutive ids (which share an operand panel) would land on different
  // per-XCD L2s; give each XCD a contiguous chunk instead (bijective form).
  int tile_n, tile_m2;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (use_swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    tile_n = bid % gridDim.x;
    tile_m2 = bid / gridDim.x;
  }
  const int m0 = tile_m2 * BM_, n0 = tile_n * BN_;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int wm0 = (wid / WAVES_N) * WM;
  const int wn0 = (wid % WAVES_N) * WN;

  f32x4 acc[MF][NF] = {};

  const long long lda = gs.lda ? gs.lda : (TA ? M : K);
  const long long ldb = gs.ldb ? gs.ldb : (TB ? K : N);
  const long long ldc = gs.ldc ? gs.ldc : N;
  // glds eligibility (bf16, full tile, 16-B-aligned rows)
  const bool glds_a = sizeof(T) == 2 && !TA && (m0 + BM_ <= M) && (lda % 8 == 0);
  const bool glds_b = sizeof(T) == 2 && TB && (n0 + BN_ <= N) && (ldb % 8 == 0);

  auto stage = [&](int buf, int k0) {
    const bool k_full = (k0 + BK <= K);
    if (TA) {
      stage_transposed<T, BM_>(a_lds(buf), A, lda, m0, k0, M, K);
    } else if (glds_a && k_full) {
      if constexpr (sizeof(T) == 2)
        stage_direct_glds((bf16*)a_lds(buf), (const bf16*)A, lda, m0, k0, BM_);
    } else {
      stage_direct<T, BM_>(a_lds(buf), A, lda, m0, k0, M, K);
    }
    if (TB) {
      if (glds_b && k_full) {
        if constexpr (sizeof(T) == 2)
          stage_direct_glds((bf16*)b_lds(buf), (const bf16*)B, ldb, n0, k0, BN_);
      } else {
        stage_direct<T, BN_>(b_lds(buf), B, ldb, n0, k0, N, K);
      }
    } else {
      stage_transposed<T, BN_>(b_lds(buf), B, ldb, n0, k0, N, K);
    }
  };

  // SPLITK: this block covers K slice [k_begin, k_end)
  const int k_begin = SPLITK ? blockIdx.z * k_chunk : 0;
  const int k_end = SPLITK ? min(k_begin + k_chunk, K) : K;
  const int nk = (k_end - k_begin + BK - 1) / BK;
  if (nk <= 0) return;
  stage(0, k_begin);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) stage(cur ^ 1, k_begin + (kt + 1) * BK);
    gemm_mma<T, MF, NF>(a_lds(cur), b_lds(cur), acc, lane, wm0, wn0);
    __syncthreads();
  }

  // Epilogue. C/D fragment map: col = lane&15, row = (lane>>4)*4 + r.
  const int col_in_frag = lane & 15;
  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm0 + mf * FRAG + row_base + r;
        int col = n0 + wn0 + nf * FRAG + col_in_frag;
        if (row < M && col < N) {
          long long off = (long long)row * ldc + col;
          float v = alpha * acc[mf][nf][r];
          if (SPLITK) {
            // fp32 atomic accumulation across K slices (TOUT = float)
            if (gridDim.z == 1)
              C[off] = (TOUT)v;
            else
              atomicAdd((float*)&C[off], v);
          } else {
            if (beta != 0.f) v = fmaf(beta, (float)C[off], v);
            if (bias != nullptr) v += bias[col];
            if (RELU) v = fmaxf(v, 0.f);
            C[off] = (TOUT)v;
          }
        }
      }
    }
  }
}

// Instantiations used by bindings.cpp. Layouts: fwd(0,1), dgrad(0,0),
// wgrad(1,0); each with bf16 and f32 compute; wgrad also with f32 out.
#define INST_GEMM(T, TOUT, TA, TB, RELU)                                     \
  template __global__ void gemm_kernel<T, TOUT, TA, TB, RELU, 128, 128, 2, 2>( \
      const T*, const T*, TOUT*, const float*, int, int, int, float, float,  \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, TOUT, TA, TB, RELU, 128, 64, 4, 1>( \
      const T*, const T*, TOUT*, const float*, int, int, int, float, float,  \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, TOUT, TA, TB, RELU, 64, 128, 1, 4>( \
      const T*, const T*, TOUT*, const float*, int, int, int, float, float,  \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, TOUT, TA, TB, RELU, 128, 32, 4, 1>( \
      const T*, const T*, TOUT*, const float*, int, int, int, float, float,  \
      long long, long long, long long, int, int, int, GemmStrides);

// split-K variants (fp32 accumulation; layouts NT and NN; all geometries)
#define INST_GEMM_SPLITK(T, TA, TB)                                          \
  template __global__ void gemm_kernel<T, float, TA, TB, false, 128, 128, 2, 2, true>( \
      const T*, const T*, float*, const float*, int, int, int, float, float, \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, float, TA, TB, false, 128, 64, 4, 1, true>( \
      const T*, const T*, float*, const float*, int, int, int, float, float, \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, float, TA, TB, false, 64, 128, 1, 4, true>( \
      const T*, const T*, float*, const float*, int, int, int, float, float, \
      long long, long long, long long, int, int, int, GemmStrides);                                 \
  template __global__ void gemm_kernel<T, float, TA, TB, false, 128, 32, 4, 1, true>( \
      const T*, const T*, float*, const float*, int, int, int, float, float, \
      long long, long long, long long, int, int, int, GemmStrides);

INST_GEMM_SPLITK(bf16, false, true)
INST_GEMM_SPLITK(bf16, false, false)
INST_GEMM_SPLITK(float, false, true)
INST_GEMM_SPLITK(float, false, false)

INST_GEMM(bf16, bf16, false, true, false)
INST_GEMM(bf16, bf16, false, true, true)
INST_GEMM(bf16, bf16, false, false, false)
INST_GEMM(bf16, bf16, true, false, false)
INST_GEMM(bf16, float, true, false, false)
INST_GEMM(float, float, false, true, false)
INST_GEMM(float, float, false, true, true)
INST_GEMM(float, float, false, false, false)
INST_GEMM(float, float, true, false, false)

// ---- launchers -------------------------------------------------------------
#include "launchers.h"

void launch_gemm_batched(bool in_bf16, bool out_f32, int layout, bool relu,
                 const void* A, const void* B, void* C, const float* bias,
                 int M, int N, int K, float alpha, float beta, int nbatch,
                 long long strideA, long long strideB, long long strideC,
                 hipStream_t s, int b_group, GemmStrides gs) {
  const bool strided = gs.heads > 1 || gs.lda || gs.ldb || gs.ldc;
  // Tile geometry: prefer 128x128; when that grid underfills the chip
  // (< ~1.5 blocks/CU), halve the narrower output dim's tile.
  long long tiles128 = ((long long)(M + 127) / 128) * ((N + 127) / 128) * nbatch;
  int geom = 0;
  if (N <= 32) geom = 3;           // 128x32 (4x1 waves) — skinny adapters
  else if (tiles128 < 384) {
    if (N <= M) geom = 1;          // 128x64 (4x1 waves)
    else geom = 2;                 // 64x128 (1x4 waves)
  }
  const int bm = geom == 2 ? 64 : 128;
  const int bn = geom == 1 ? 64 : (geom == 3 ? 32 : 128);
  dim3 grid((N + bn - 1) / bn, (M + bm - 1) / bm, nbatch);
  dim3 block(kBlock);
  // XCD swizzle only when the operands exceed the 256 MiB L3 (T1: costs a
  // few % when L3-resident, pays ~10% when HBM-bound)
  const long long op_bytes =
      ((long long)M * K + (long long)N * K) * (in_bf16 ? 2 : 4) * nbatch;
  const int use_swz = op_bytes > (200LL << 20) ? 1 : 0;
  // Deep-pipelined 256x256 8-wave kernel for big bf16 NT tiles (gemm8.hip).
  // Ragged M (e.g. token counts): run the 256-aligned row block on the
  // deep kernel and the remainder rows on the 2-phase kernel below.
  if (in_bf16 && !out_f32 && layout == 0 && !relu && nbatch == 1 &&
      beta == 0.f && !strided) {   // fp32 bias is fused in the 8-phase epilogue
    const int m_main = (M / 256) * 256;
    if (m_main == M) {
      if (launch_gemm_nt_8ph(A, B, C, bias, M, N, K, alpha, use_swz, s))
        return;
    } else if (m_main > 0 && launch_gemm_nt_8ph(A, B, C, bias, m_main, N, K,
                                                alpha, use_swz, s)) {
      const bf16* A_rem = (const bf16*)A + (long long)m_main * K;
      bf16* C_rem = (bf16*)C + (long long)m_main * N;
      launch_gemm_batched(true, false, 0, false, A_rem, B, C_rem, bias,
                          M - m_main, N, K, alpha, 0.f, 1, 0, 0, 0, s, 1);
      return;
    }
  }
  #define GEMM_CALL(T, TOUT, TA, TB, RELU)                                    \
    do {                                                                      \
      if (geom == 1)                                                          \
        hipLaunchKernelGGL((gemm_kernel<T, TOUT, TA, TB, RELU, 128, 64, 4, 1>),\
                           grid, block, 0, s, (const T*)A, (const T*)B,       \
                           (TOUT*)C, bias, M, N, K, alpha, beta, strideA,     \
                           strideB, strideC, 0, use_swz, b_group, gs);         \
      else if (geom == 2)                                                     \
        hipLaunchKernelGGL((gemm_kernel<T, TOUT, TA, TB, RELU, 64, 128, 1, 4>),\
                           grid, block, 0, s, (const T*)A, (const T*)B,       \
                           (TOUT*)C, bias, M, N, K, alpha, beta, strideA,     \
                           strideB, strideC, 0, use_swz, b_group, gs);         \
      else if (geom == 3)                                                     \
        hipLaunchKernelGGL((gemm_kernel<T, TOUT, TA, TB, RELU, 128, 32, 4, 1>),\
                           grid, block, 0, s, (const T*)A, (const T*)B,       \
                           (TOUT*)C, bias, M, N, K, alpha, beta, strideA,     \
                           strideB, strideC, 0, use_swz, b_group, gs);         \
      else                                                                    \
        hipLaunchKernelGGL((gemm_kernel<T, TOUT, TA, TB, RELU, 128, 128, 2, 2>),\
                           grid, block, 0, s, (const T*)A, (const T*)B,       \
                           (TOUT*)C, bias, M, N, K, alpha, beta, strideA,     \
                           strideB, strideC, 0, use_swz, b_group, gs);         \
    } while (0)
  if (in_bf16) {
    if (layout == 0) {          // NT: fwd
      if (relu) GEMM_CALL(bf16, bf16, false, true, true);
      else      GEMM_CALL(bf16, bf16, false, true, false);
    } else if (layout == 1) {   // NN: dgrad
      GEMM_CALL(bf16, bf16, false, false, false);
    } else {                    // TN: wgrad
      if (out_f32) GEMM_CALL(bf16, float, true, false, false);
      else         GEMM_CALL(bf16, bf16, true, false, false);
    }
  } else {
    if (layout == 0) {
      if (relu) GEMM_CALL(float, float, false, true, true);
      else      GEMM_CALL(float, float, false, true, false);
    } else if (layout == 1) {
      GEMM_CALL(float, float, false, false, false);
    } else {
      GEMM_CALL(float, float, true, false, false);
    }
  }
  #undef GEMM_CALL
}

void launch_gemm(bool in_bf16, bool out_f32, int layout, bool relu,
                 const void* A, const void* B, void* C, const float* bias,
                 int M, int N, int K, float alpha, float beta, hipStream_t s) {
  launch_gemm_batched(in_bf16, out_f32, layout, relu, A, B, C, bias, M, N, K,
                      alpha, beta, 1, 0, 0, 0, s);
}

void launch_gemm_splitk(bool in_bf16, int layout, const void* A,
                        const void* B, float* C, int M, int N, int K,
                        hipStream_t s) {
  // layout 0 = NT, 1 = NN. Geometry as the dense launcher, then split K to
  // fill ~2 blocks/CU.
  int geom = 0;
  long long tiles128 = ((long long)(M + 127) / 128) * ((N + 127) / 128);
  // 128x128 has 2x the arithmetic intensity of the narrow tiles and the
  // K-split supplies the grid fill, so it wins whenever the chip can be
  // covered at all (measured +45..100% on the BERT wgrad shapes, BERT
  // end-to-end +12%); narrow geometries only when even max splits cannot
  // reach 256 blocks, 128x32 for the skinny-N LoRA shapes.
  if (N <= 32) geom = 3;
  else if (tiles128 * ((K + BK - 1) / BK) < 256) geom = (N <= M) ? 1 : 2;
  // A/B override: BATON_SK_GEOM forces a split-K tile geometry (perf runs)
  {
    static int force = [] {
      const char* e = std::getenv("BATON_SK_GEOM");
      return e ? std::atoi(e) : -1;
    }();
    if (force >= 0 && !(force != 3 && N <= 32)) geom = force;
  }
  const int bm = geom == 2 ? 64 : 128;
  const int bn = geom == 1 ? 64 : (geom == 3 ? 32 : 128);
  long long tiles = ((long long)(M + bm - 1) / bm) * ((N + bn - 1) / bn);
  int splits = (int)(512 / (tiles > 0 ? tiles : 1));
  if (splits < 1) splits = 1;
  int max_splits = (K + BK - 1) / BK;
  if (splits > max_splits) splits = max_splits;
  int k_chunk = ((K + splits - 1) / splits + BK - 1) / BK * BK;
  splits = (K + k_chunk - 1) / k_chunk;
  dim3 grid((N + bn - 1) / bn, (M + bm - 1) / bm, splits);
  #define SK_CALL(T, TA, TB)                                                   \
    do {                                                                       \
      if (geom == 1)                                                           \
        hipLaunchKernelGGL((gemm_kernel<T, float, TA, TB, false, 128, 64, 4, 1, true>), \
                           grid, dim3(kBlock), 0, s, (const T*)A, (const T*)B, \
                           C, nullptr, M, N, K, 1.f, 0.f, 0, 0, 0, k_chunk, 0, 1);   \
      else if (geom == 2)                                                      \
        hipLaunchKernelGGL((gemm_kernel<T, float, TA, TB, false, 64, 128, 1, 4, true>), \
                           grid, dim3(kBlock), 0, s, (const T*)A, (const T*)B, \
                           C, nullptr, M, N, K, 1.f, 0.f, 0, 0, 0, k_chunk, 0, 1);   \
      else if (geom == 3)                                                      \
        hipLaunchKernelGGL((gemm_kernel<T, float, TA, TB, false, 128, 32, 4, 1, true>), \
                           grid, dim3(kBlock), 0, s, (const T*)A, (const T*)B, \
                           C, nullptr, M, N, K, 1.f, 0.f, 0, 0, 0, k_chunk, 0, 1);   \
      else                                                                     \
        hipLaunchKernelGGL((gemm_kernel<T, float, TA, TB, false, 128, 128, 2, 2, true>), \
                           grid, dim3(kBlock), 0, s, (const T*)A, (const T*)B, \
                           C, nullptr, M, N, K, 1.f, 0.f, 0, 0, 0, k_chunk, 0, 1);   \
    } while (0)
  if (in_bf16) {
    if (layout == 0) SK_CALL(bf16, false, true);
    else if (layout == 2) SK_CALL(bf16, true, false);  // native TN (wgrads)
    else SK_CALL(bf16, false, false);
  } else {
    if (layout == 0) SK_CALL(float, false, true);
    else if (layout == 2) SK_CALL(float, true, false);
    else SK_CALL(float, false, false);
  }
  #undef SK_CALL
}
