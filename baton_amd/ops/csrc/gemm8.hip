// Deep-pipelined NT GEMM (gfx950): 256x256 tile, 8 waves, glds staging
// with COUNTED vmcnt across RAW barriers — the "8-phase" structure of the
// CDNA4 guide (§5.5 T3+T4): global_load_lds stays in flight across
// barriers instead of draining at every k-step, which is what caps the
// simple 2-phase pipeline (~650 TF) well below the chip's reach.
//
// Schedule (constructed for this kernel; bf16 NT, full tiles only):
//   * K walks in 32-wide "k-halves"; LDS holds FOUR slot-pairs
//     (A[256][32] + B[256][32] = 32 KiB each, 128 KiB total), rotating
//     slot = kh & 3;
//   * each k-half runs TWO phases: phase 0 computes nf-pair {0,1}
//     (8 mf x 2 nf = 16 MFMAs/wave), phase 1 computes nf-pair {2,3}
//     reusing the A fragments read in phase 0;
//   * phase 0 of k-half kh issues the glds for A(kh+3), phase 1 for
//     B(kh+3) — 2 x global_load_lds_dwordx4 per thread per phase —
//     into the slot freed at the end of k-half kh-1;
//   * ONE counted wait per k-half, before phase 0's ds_reads: everything
//     issued after B(kh) may stay in flight — 5 slot-stages x 2 glds =
//     s_waitcnt vmcnt(10) in steady state, tightened near the K tail;
//   * barriers are RAW s_barrier (+ explicit lgkmcnt(0) before the MFMA
//     cluster): __syncthreads() would emit vmcnt(0) while a glds is in
//     flight and drain the pipeline (guide §5 'Pipelining across
//     barriers');
//   * s_setprio(1) around each MFMA cluster (T5 — pays exactly on this
//     phase-split structure);
//   * single __shared__ object (multiple LDS objects make hipcc emit
//     vmcnt(0) before every k-step's first ds_read — guide §5 trap (a));
//   * LDS image identical to gemm.hip: linear 64-B rows, 16-B slot XOR
//     swizzle on the glds SOURCE address and the ds_read offsets.
//
// Eligibility (launcher-checked): bf16, NT, M % 256 == 0, N % 256 == 0,
// K % 64 == 0 (=> rows 16-B aligned), no bias/beta/relu. Everything else
// takes the 2-phase kernel in gemm.hip.
#include "common.h"
#include <cstdlib>
#include <type_traits>

namespace g8 {

constexpr int KH = 32;              // k-half width (one MFMA K)
constexpr int TM = 256, TN = 256;   // C tile
constexpr int THREADS = 512;        // 8 waves (2 M x 4 N)
constexpr int SLOT = 256 * KH;      // elements per operand slot

// slot-swizzled element offset within a [256][32] bf16 image (64-B rows)
DEVINL int soff(int row, int col) {
  const int sl = col >> 3;
  return row * KH + ((sl ^ ((row >> 2) & 3)) << 3) + (col & 7);
}

// stage one operand slot (256 rows x 32 k) via glds; 2 dwordx4 per thread
DEVINL void stage_slot(bf16* __restrict__ lds, const bf16* __restrict__ src,
                       long long ld, int k0) {
  const int t = threadIdx.x;
  const int w = t >> 6;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int idx = p * THREADS + t;
    const int row = idx >> 2;
    const int psl = idx & 3;
    const int lsl = psl ^ ((row >> 2) & 3);
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        src + (long long)row * ld + k0 + lsl * 8);
    auto l = (__attribute__((address_space(3))) unsigned int*)(
        lds + (long long)(p * THREADS + w * 64) * 8);
    __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  }
}

// counted wait: allow `n` vector-memory ops to stay in flight
DEVINL void vmwait(int n) {
  switch (n) {
    case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
    case 2: asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); break;
    case 4: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
    case 6: asm volatile("s_waitcnt vmcnt(6)" ::: "memory"); break;
    case 8: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
    default: asm volatile("s_waitcnt vmcnt(10)" ::: "memory"); break;
  }
}

// SCHED selects the in-loop schedule for A/B experiments (BATON_G8_SCHED):
//   0 = two phases per k-half, b-frags {2,3} read in phase 1 (r1 shipped)
//   1 = all 12 fragment reads up front, phase 1 = stage B + MFMA only
//   2 = fully merged: both stages issued before the wait, one 32-MFMA
//       cluster per k-half
template <bool SPLITK, typename TOUT, int SCHED = 0>
__global__ __launch_bounds__(THREADS) void gemm_nt_8ph_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    TOUT* __restrict__ C, const float* __restrict__ bias, int M, int N, int K,
    float alpha, int use_swz, int k_chunk) {
  // ONE shared object: 4 slot-pairs [A | B]
  __shared__ bf16 lds[4 * 2 * SLOT];
  auto a_slot = [&](int s) -> bf16* { return lds + s * 2 * SLOT; };
  auto b_slot = [&](int s) -> bf16* { return lds + s * 2 * SLOT + SLOT; };

  int tile_n, tile_m;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (use_swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    tile_n = bid % gridDim.x;
    tile_m = bid / gridDim.x;
  }
  const bf16* Atile = A + (long long)tile_m * TM * K;   // lda = K
  const bf16* Btile = B + (long long)tile_n * TN * K;   // ldb = K
  const int m0 = tile_m * TM, n0 = tile_n * TN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm0 = (wid >> 2) * 128;     // wave rows: 128*wm
  const int wn0 = (wid & 3) * 64;       // wave cols: 64*wn

  f32x4 acc[8][4] = {};                 // 128 VGPRs of accumulator

  // SPLITK: this block covers K slice [kb, kb + k_chunk)
  const int kb = SPLITK ? blockIdx.z * k_chunk : 0;
  const int ke = SPLITK ? min(kb + k_chunk, K) : K;
  const int nkh = (ke - kb) / KH;
  if (nkh <= 0) return;
  // prologue: stage k-halves 0..2 (or fewer)
  const int pro = nkh < 3 ? nkh : 3;
  for (int j = 0; j < pro; ++j) {
    stage_slot(a_slot(j & 3), Atile, K, kb + j * KH);
    stage_slot(b_slot(j & 3), Btile, K, kb + j * KH);
  }

  const int arow = lane & 15;           // fragment row within 16
  const int kfrag = (lane >> 4) * 8;    // fragment k offset within 32

  for (int kh = 0; kh < nkh; ++kh) {
    const int s = kh & 3;
    const bf16* As = a_slot(s);
    const bf16* Bs = b_slot(s);

    // ---- phase 0: issue A(kh+3) (and B under SCHED 2), wait, MFMA
    if (kh + 3 < nkh) stage_slot(a_slot((kh + 3) & 3), Atile, K, kb + (kh + 3) * KH);
    if (SCHED == 2 && kh + 3 < nkh)
      stage_slot(b_slot((kh + 3) & 3), Btile, K, kb + (kh + 3) * KH);
    {
      // outstanding allowed = stages issued after B(kh):
      //   full slot-pairs for kh+1..min(kh+2, nkh-1)  (2 stages each)
      //   + this phase's A(kh+3) (+B under SCHED 2) if it exists
      int ahead = 0;
      if (kh + 1 <= nkh - 1) ++ahead;
      if (kh + 2 <= nkh - 1) ++ahead;
      int stages = 2 * ahead +
                   (kh + 3 <= nkh - 1 ? (SCHED == 2 ? 2 : 1) : 0);
      vmwait(2 * stages);
    }
    __builtin_amdgcn_s_barrier();       // every wave's slot data visible

    s16x8 a_frag[8], b_frag[4];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &As[soff(wm0 + mf * 16 + arow, kfrag)]);
#pragma unroll
    for (int nf = 0; nf < (SCHED == 0 ? 2 : 4); ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &Bs[soff(wn0 + nf * 16 + arow, kfrag)]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 0; nf < (SCHED == 2 ? 4 : 2); ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- phase 1 (SCHED 0/1): issue B(kh+3), MFMA nf-pair {2,3}
    if (SCHED != 2) {
      if (kh + 3 < nkh)
        stage_slot(b_slot((kh + 3) & 3), Btile, K, kb + (kh + 3) * KH);
      if (SCHED == 0) {
#pragma unroll
        for (int nf = 2; nf < 4; ++nf)
          b_frag[nf] = *reinterpret_cast<const s16x8*>(
              &Bs[soff(wn0 + nf * 16 + arow, kfrag)]);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 2; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();       // slot s free for staging at kh+1
  }

  // drain every outstanding glds before the epilogue reuses registers
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int col_in_frag = lane & 15;
  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
    const int col = n0 + wn0 + nf * 16 + col_in_frag;
    // fp32 bias fused into the epilogue (null on the split-K/atomic path —
    // a per-slice add would apply it gridDim.z times)
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm0 + mf * 16 + row_base + r;
        if (SPLITK)
          // per-slice slab (plain stores; a tiny reduce kernel sums the
          // slices) — fp32 atomics at this tile size quadruple the
          // epilogue's atomic volume and measured slower (r1 profiles)
          C[(long long)blockIdx.z * M * N + (long long)row * N + col] =
              (TOUT)(alpha * acc[mf][nf][r]);
        else
          C[(long long)row * N + col] =
              (TOUT)(alpha * acc[mf][nf][r] + bv);
      }
  }
}

}  // namespace g8

// ---- g9: the guide-template fine-phase schedule ----------------------------
// 256x256 tile, BK=64 K-tiles staged as FOUR [128][64] half-tiles, one
// half-tile glds-staged per phase, 4 quadrant phases per K-tile (16 MFMA
// each over the full K=64). The ds reads for phase p+1's quadrant issue at
// the TOP of phase p, so they sit beside the partner waves' MFMA; ONE
// counted vmcnt(6) per K-tile keeps 3 half-tiles in flight across the
// barriers. Swizzle: st_16x32 (col bit4 ^ row bit2) on the [128][64]
// images, applied on the glds SOURCE slot and the read address.
//
// Schedule invariants (derived; see quadrant walk below):
//   * quadrants (0,0)->(0,1)->(1,1)->(1,0): each phase transition changes
//     ONE operand half, so a phase issues 4, 8 or 12 ds_read_b128s;
//   * a wave re-reads its A image at phases 4t-1 and 4t+1 and its B image
//     at 4t-1 .. 4t+2 (the quadrant walk stays inside the wave's own
//     halves), so slot depths are ASYMMETRIC: A 2-deep, B 3-deep (10 x
//     16 KiB images = the full 160 KiB LDS). With per-tile stage order
//     [B0, B1, A0, A1] and the stage pointer 8 halves ahead, every slot
//     rewrite lands at least one barrier after its previous occupant's
//     last read, and the per-tile vmcnt(6) at quadrant 2 lands every half
//     of tile t+1 before its first read at phase 4t+3;
//   * the LAST in-loop wait drains to 0 (the tail has no younger stages
//     for the count to push against).
namespace g9 {

constexpr int TM = 256, TN = 256, BK = 64;
constexpr int THREADS = 512;
constexpr int HALF = 128 * BK;

DEVINL int e9(int row, int col) {
  return row * BK + (col ^ (((row >> 2) & 1) << 4));
}

DEVINL void vmwait6or0(bool six) {
  if (six) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

DEVINL void stage_half(bf16* __restrict__ lds, const bf16* __restrict__ src,
                       long long ld, int r0, int k0) {
  const int t = threadIdx.x;
  const int w = t >> 6;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int idx = p * THREADS + t;
    const int row = idx >> 3;
    const int slot = idx & 7;
    const int slotp = slot ^ (((row >> 2) & 1) << 1);
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        src + (long long)(r0 + row) * ld + k0 + slotp * 8);
    auto l = (__attribute__((address_space(3))) unsigned int*)(
        lds + (long long)(p * THREADS + w * 64) * 8);
    __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  }
}

// NOWAIT: skip the explicit lgkmcnt(0) before each MFMA cluster and let
// hipcc's own counted lgkm ladders keep the NEXT phase's just-issued
// reads in flight under the current cluster (the explicit 0-drain
// serializes them).
template <bool NOWAIT, bool SPLITK = false, typename TOUT = bf16>
__global__ __launch_bounds__(THREADS) void gemm_nt_g9_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    TOUT* __restrict__ C, const float* __restrict__ bias, int M, int N, int K,
    float alpha, int use_swz, int k_chunk) {
  // 10 half-slots: A [mhalf][t%2], B [mhalf][t%3]
  __shared__ bf16 lds[10 * HALF];
  auto slot = [&](int op, int mh, int t) -> bf16* {
    const int idx = op == 0 ? mh * 2 + (t & 1) : 4 + mh * 3 + (t % 3);
    return lds + idx * HALF;
  };

  int tile_n, tile_m;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (use_swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    tile_n = bid % gridDim.x;
    tile_m = bid / gridDim.x;
  }
  const bf16* Atile = A + (long long)tile_m * TM * K;
  const bf16* Btile = B + (long long)tile_n * TN * K;
  const int m0 = tile_m * TM, n0 = tile_n * TN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wmh = wid >> 2;             // wave's A half (rows wmh*128..)
  const int wn0 = (wid & 3) * 64;       // wave cols within 256
  const int wnh = wn0 >> 7;             // wave's B half image
  const int wnr = wn0 & 127;            // row offset inside that image

  const int arow = lane & 15;
  const int kfrag = (lane >> 4) * 8;

  f32x4 acc[8][4] = {};

  const int kb = SPLITK ? blockIdx.z * k_chunk : 0;
  const int ntile = (SPLITK ? k_chunk : K) / BK;
  const int nphase = ntile * 4;
  // per-tile stage order [B0, B1, A0, A1] (A must land in phases >= 4t+2
  // of its slot's previous occupant — see header invariants)
  auto stage_h = [&](int h) {
    if (h >= nphase) return;
    const int t = h >> 2, j = h & 3;
    if (j == 0) stage_half(slot(1, 0, t), Btile, K, 0, kb + t * BK);
    else if (j == 1) stage_half(slot(1, 1, t), Btile, K, 128, kb + t * BK);
    else if (j == 2) stage_half(slot(0, 0, t), Atile, K, 0, kb + t * BK);
    else stage_half(slot(0, 1, t), Atile, K, 128, kb + t * BK);
  };
  const int pro = nphase < 8 ? nphase : 8;
  for (int h = 0; h < pro; ++h) stage_h(h);
  if (ntile >= 2)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile 0 landed
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  s16x8 afr[2][4][2];                   // [set][mf][kstep]
  s16x8 bfr[2][2][2];                   // [set][nf][kstep]

  // reads for phase 0 (quadrant 0,0 of tile 0)
  {
    const bf16* As = slot(0, wmh, 0);
    const bf16* Bs = slot(1, wnh, 0);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[0][mf][ks] = *reinterpret_cast<const s16x8*>(
            &As[e9(mf * 16 + arow, ks * 32 + kfrag)]);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[0][nf][ks] = *reinterpret_cast<const s16x8*>(
            &Bs[e9(wnr + nf * 16 + arow, ks * 32 + kfrag)]);
  }

  // quadrant loop unrolled so every acc/frag index is compile-time
  // (runtime-indexed ext_vector arrays spill to scratch — guide rule 20).
  // A changes twice per tile, so its set usage is period-4: ause=(q>=2).
  // B changes THREE times per tile (q0->q1, q2->q3, q3->q0'), an odd
  // count, so its double-buffer parity alternates per TILE: the B
  // instance live at phase q of tile t is the (3t + {0,1,1,2}[q])-th,
  // giving buse=(t+{0,1,1,2}[q])&1 — even tiles walk 0,1,1,0 and odd
  // tiles 1,0,0,1. The tile body is a lambda over a compile-time parity
  // so every frag index stays compile-time: the loop walks full tile
  // PAIRS and an odd ntile gets an explicit even-parity tail (an early
  // `break` inside an unrolled sub-loop defeats the unroll and spills —
  // measured 1136 B/lane scratch).
  // Reads at phase q target the next instance's set (buse^1).
  auto tile_body = [&](int tile, auto subc) {
    constexpr int sub = decltype(subc)::value;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int p = tile * 4 + q;
      constexpr int QMH[4] = {0, 0, 1, 1};
      constexpr int QNH[4] = {0, 1, 1, 0};
      constexpr int BQ[4] = {0, 1, 1, 2};
      const int qmh = QMH[q], qnh = QNH[q];
      const int qn = (q + 1) & 3;
      const int nmh = QMH[qn], nnh = QNH[qn];
      const int tn = q == 3 ? tile + 1 : tile;
      const int ause = (q >= 2) ? 1 : 0;
      const int buse = (sub + BQ[q]) & 1;

      // ---- issue next phase's ds reads + this phase's half-tile stage
      if (p + 1 < nphase) {
        if (nmh != qmh || qn == 0) {
          const bf16* As = slot(0, wmh, tn);
          const int rbase = nmh * 64;
#pragma unroll
          for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              afr[ause ^ 1][mf][ks] = *reinterpret_cast<const s16x8*>(
                  &As[e9(rbase + mf * 16 + arow, ks * 32 + kfrag)]);
        }
        if (nnh != qnh || qn == 0) {
          const bf16* Bs = slot(1, wnh, tn);
          const int cbase = nnh * 32;
#pragma unroll
          for (int nf = 0; nf < 2; ++nf)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              bfr[buse ^ 1][nf][ks] = *reinterpret_cast<const s16x8*>(
                  &Bs[e9(wnr + cbase + nf * 16 + arow, ks * 32 + kfrag)]);
        }
      }
      stage_h(p + 8);

      __builtin_amdgcn_s_barrier();
      if (!NOWAIT) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          const int am = qmh * 4 + mf, bn = qnh * 2 + nf;
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[am][bn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[ause][mf][ks], bfr[buse][nf][ks], acc[am][bn], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
      if (q == 2 && ntile >= 2)
        vmwait6or0(tile < ntile - 2);
      __builtin_amdgcn_s_barrier();
    }
  };
  const int npair = ntile & ~1;
  for (int tp = 0; tp < npair; tp += 2) {
    tile_body(tp, std::integral_constant<int, 0>{});
    tile_body(tp + 1, std::integral_constant<int, 1>{});
  }
  if (ntile & 1)  // ntile odd => last tile index is even => parity 0
    tile_body(ntile - 1, std::integral_constant<int, 0>{});
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int col_in_frag = lane & 15;
  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int bn = 0; bn < 4; ++bn) {
    const int col = n0 + wn0 + (bn >> 1) * 32 + (bn & 1) * 16 + col_in_frag;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int am = 0; am < 8; ++am)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wmh * 128 + (am >> 2) * 64 + (am & 3) * 16 +
                        row_base + r;
        if (SPLITK)
          C[(long long)blockIdx.z * M * N + (long long)row * N + col] =
              (TOUT)(alpha * acc[am][bn][r]);
        else
          C[(long long)row * N + col] = (TOUT)(alpha * acc[am][bn][r] + bv);
      }
  }
}

}  // namespace g9

namespace g8 {

// sum SPLITK fp32 slabs [S, M*N] into out (bf16 or fp32), vectorized
template <typename TOUT>
__global__ __launch_bounds__(kBlock) void splitk_reduce_kernel(
    const float* __restrict__ slabs, TOUT* __restrict__ out, long long mn,
    int nslab) {
  const long long nvec = mn / 4;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    f32x4 acc = reinterpret_cast<const f32x4*>(slabs)[i];
    for (int s = 1; s < nslab; ++s) {
      const f32x4 v =
          reinterpret_cast<const f32x4*>(slabs + (long long)s * mn)[i];
#pragma unroll
      for (int q = 0; q < 4; ++q) acc[q] += v[q];
    }
    if constexpr (sizeof(TOUT) == 2) {
      ushort2 lo{f32_to_bf16(acc[0]), f32_to_bf16(acc[1])};
      ushort2 hi{f32_to_bf16(acc[2]), f32_to_bf16(acc[3])};
      reinterpret_cast<ushort2*>(out)[i * 2] = lo;
      reinterpret_cast<ushort2*>(out)[i * 2 + 1] = hi;
    } else {
      reinterpret_cast<f32x4*>(out)[i] = *reinterpret_cast<f32x4*>(&acc);
    }
  }
  for (long long i = nvec * 4 + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < mn; i += (long long)gridDim.x * blockDim.x) {
    float a = slabs[i];
    for (int s = 1; s < nslab; ++s) a += slabs[(long long)s * mn + i];
    out[i] = (TOUT)a;
  }
}

}  // namespace g8

#include "launchers.h"

bool launch_gemm_nt_8ph(const void* A, const void* B, void* C,
                        const float* bias, int M, int N, int K, float alpha,
                        int use_swz, hipStream_t s) {
  if (M % g8::TM != 0 || N % g8::TN != 0 || K % 64 != 0) return false;
  // one 8-wave block per CU: the grid must roughly cover the 256 CUs or
  // the higher-occupancy 2-phase kernel wins (measured: 554 vs 871 TF at
  // 128 blocks; at 192 blocks the 8-phase already wins — BERT fwd N=768
  // shapes, r2 A/B +1%); threshold overridable for A/B runs
  static long long min_blocks = [] {
    const char* e = std::getenv("BATON_G8_MIN_BLOCKS");
    return e ? std::atoll(e) : 160LL;
  }();
  if ((long long)(M / g8::TM) * (N / g8::TN) < min_blocks) return false;
  static int sched = [] {
    const char* e = std::getenv("BATON_G8_SCHED");
    return e ? std::atoi(e) : 4;  // g9 fine-phase, compiler-counted lgkm (r2 A/B winner)
  }();
  dim3 grid(N / g8::TN, M / g8::TM);
#define G8_CALL(SC)                                                           \
  hipLaunchKernelGGL((g8::gemm_nt_8ph_kernel<false, bf16, SC>), grid,         \
                     dim3(g8::THREADS), 0, s, (const bf16*)A, (const bf16*)B, \
                     (bf16*)C, bias, M, N, K, alpha, use_swz, 0)
  if (sched == 3 || sched == 4) {
    if (sched == 4)
      hipLaunchKernelGGL((g9::gemm_nt_g9_kernel<true>), grid,
                         dim3(g9::THREADS), 0, s, (const bf16*)A,
                         (const bf16*)B, (bf16*)C, bias, M, N, K, alpha,
                         use_swz, 0);
    else
      hipLaunchKernelGGL((g9::gemm_nt_g9_kernel<false>), grid,
                         dim3(g9::THREADS), 0, s, (const bf16*)A,
                         (const bf16*)B, (bf16*)C, bias, M, N, K, alpha,
                         use_swz, 0);
  }
  else if (sched == 1) G8_CALL(1);
  else if (sched == 2) G8_CALL(2);
  else G8_CALL(0);
#undef G8_CALL
  return true;
}

// Split-K on the 8-phase kernel: grid.z slices each write a private fp32
// slab (plain stores), then splitk_reduce sums them. For the long-K
// skinny-tile wgrad shapes (BERT dW: 9-36 tiles on 256 CUs) this replaces
// the 2-phase + fp32-atomic path. Caller picks splitk so that
// tiles * splitk covers the chip and K/splitk is a multiple of 32.
bool launch_gemm_nt_8ph_splitk(const void* A, const void* B, float* slabs,
                               void* out, bool out_bf16, int M, int N, int K,
                               int splitk, int use_swz, hipStream_t s) {
  if (M % g8::TM != 0 || N % g8::TN != 0) return false;
  const int k_chunk = K / splitk;
  if (k_chunk * splitk != K || k_chunk % 32 != 0 || k_chunk < 64) return false;
  static int sched9 = [] {
    const char* e = std::getenv("BATON_G8_SCHED");
    return e ? std::atoi(e) : 4;
  }();
  dim3 grid(N / g8::TN, M / g8::TM, splitk);
  static int g9wait = [] {
    const char* e = std::getenv("BATON_G9SPLIT_WAIT");
    return e ? std::atoi(e) : 0;
  }();
  if (sched9 >= 3 && k_chunk % 64 == 0) {
    if (g9wait)
      hipLaunchKernelGGL((g9::gemm_nt_g9_kernel<false, true, float>), grid,
                         dim3(g9::THREADS), 0, s, (const bf16*)A,
                         (const bf16*)B, slabs, nullptr, M, N, K, 1.0f,
                         use_swz, k_chunk);
    else
      hipLaunchKernelGGL((g9::gemm_nt_g9_kernel<true, true, float>), grid,
                         dim3(g9::THREADS), 0, s, (const bf16*)A,
                         (const bf16*)B, slabs, nullptr, M, N, K, 1.0f,
                         use_swz, k_chunk);
  }
  else
    hipLaunchKernelGGL((g8::gemm_nt_8ph_kernel<true, float>), grid,
                       dim3(g8::THREADS), 0, s, (const bf16*)A, (const bf16*)B,
                       slabs, nullptr, M, N, K, 1.0f, use_swz, k_chunk);
  const long long mn = (long long)M * N;
  const int rgrid = elementwise_grid(mn / 4 + 1);
  if (out_bf16)
    hipLaunchKernelGGL(g8::splitk_reduce_kernel<bf16>, dim3(rgrid),
                       dim3(kBlock), 0, s, slabs, (bf16*)out, mn, splitk);
  else
    hipLaunchKernelGGL(g8::splitk_reduce_kernel<float>, dim3(rgrid),
                       dim3(kBlock), 0, s, slabs, (float*)out, mn, splitk);
  return true;
}

