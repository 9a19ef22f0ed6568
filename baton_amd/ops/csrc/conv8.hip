// Deep-pipelined implicit-GEMM conv (gfx950): 256x256 tile, 8 waves,
// ALL-glds staging with counted vmcnt across raw barriers — the gemm8
// structure applied to conv fwd/dgrad (ROUND2 plan item 2).
//
// The 2-phase conv kernels (conv.hip) hit the ~650 TF structural ceiling:
// their __syncthreads drains the in-flight global_load_lds every k-step
// (guide §5 'Pipelining across barriers'). The fix needs BOTH operands on
// glds so the counted s_waitcnt vmcnt(N) governs the whole pipeline
// (mixing ordinary loads with glds makes hipcc emit vmcnt(0) at each use
// — guide §5 trap (b)). The conv A operand is a GATHER with zero-filled
// out-of-bounds taps, which glds cannot fabricate from registers — so
// out-of-bounds lanes point their SOURCE address at a 16-byte zero page
// in global memory (L2-resident, broadcast) and the destination stays
// lane-linear. The XOR slot swizzle rides on the source k-offset exactly
// as in gemm8.
//
// Eligibility (launcher-checked): bf16, stride 1 (dgrad) / any (fwd),
// channels % 32 == 0, pixels % 256 == 0, Cout (fwd) / Cin (dgrad) % 256
// == 0, K % 64 == 0. Everything else keeps the 2-phase kernels.
#include "common.h"

namespace c8 {

constexpr int KHALF = 32;
constexpr int TM = 256, TN = 256;
constexpr int THREADS = 512;
constexpr int SLOT = 256 * KHALF;

__device__ __align__(16) const unsigned int g_zero16[4] = {0, 0, 0, 0};

struct Shape {
  int N, H, W, Cin, Cout, KH, KW, stride, pad, HO, WO, swz;
};

// slot-swizzled element offset within a [256][32] bf16 image (64-B rows)
DEVINL int soff(int row, int col) {
  const int sl = col >> 3;
  return row * KHALF + ((sl ^ ((row >> 2) & 3)) << 3) + (col & 7);
}

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

// one glds 16-B piece: dest lane-linear, source per-lane (zero page if oob)
DEVINL void glds16(const bf16* src_or_zero, bf16* lds_dst_base, int piece) {
  const int w = threadIdx.x >> 6;
  auto g = (const __attribute__((address_space(1))) unsigned int*)src_or_zero;
  auto l = (__attribute__((address_space(3))) unsigned int*)(
      lds_dst_base + (long long)(piece * THREADS + w * 64) * 8);
  __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
}

// B operand (weights / w_t): dense row-major k-contiguous, rows = filters.
// Identical to gemm8's stage_slot.
DEVINL void stage_B(bf16* __restrict__ lds, const bf16* __restrict__ src,
                    long long ld, int n0, int k0) {
  const int t = threadIdx.x;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int idx = p * THREADS + t;
    const int row = idx >> 2;
    const int psl = idx & 3;
    const int lsl = psl ^ ((row >> 2) & 3);
    glds16(src + (long long)(n0 + row) * ld + k0 + lsl * 8, lds, p);
  }
}

// Forward A stager: rows = output pixels, k = (kh,kw,ci), Cin % 32 == 0.
// Pixel decode hoisted to init; tap state advances incrementally.
struct FwdA {
  int n[2], ho[2], wo[2];     // per-pass pixel coords (row fixed per pass)
  bool rok[2];
  int lsl;                    // this thread's swizzled k slot (0..3)
  int ci, kw, kh;             // tap cursor

  DEVINL void init(const Shape& sh, int m0, long long Mtot) {
    const int t = threadIdx.x;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = p * THREADS + t;
      const int row = idx >> 2;
      const long long m = m0 + row;
      wo[p] = (int)(m % sh.WO);
      const long long q = m / sh.WO;
      ho[p] = (int)(q % sh.HO);
      n[p] = (int)(q / sh.HO);
      rok[p] = m < Mtot;
    }
    const int psl = (t & 3);
    // lsl depends on row parity; rows differ per pass — store psl, apply
    // the XOR per pass in stage() (row>>2 & 3 differs by pass)
    lsl = psl;
    ci = 0; kw = 0; kh = 0;
  }

  DEVINL void stage(bf16* __restrict__ lds, const bf16* __restrict__ x,
                    const Shape& sh) {
    const int t = threadIdx.x;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = p * THREADS + t;
      const int row = idx >> 2;
      const int sl = lsl ^ ((row >> 2) & 3);
      const int hi = ho[p] * sh.stride - sh.pad + kh;
      const int wi = wo[p] * sh.stride - sh.pad + kw;
      const bool ok = rok[p] && hi >= 0 && hi < sh.H && wi >= 0 && wi < sh.W;
      const bf16* src = ok
          ? x + (((long long)n[p] * sh.H + hi) * sh.W + wi) * sh.Cin + ci +
                sl * 8
          : (const bf16*)g_zero16;
      glds16(src, lds, p);
    }
    ci += KHALF;
    if (ci >= sh.Cin) {
      ci = 0;
      if (++kw == sh.KW) { kw = 0; ++kh; }
    }
  }
};

// Dgrad A stager (stride 1): rows = input pixels, k = (kh,kw,co),
// Cout % 32 == 0. dy pixel = (hi + pad - kh, wi + pad - kw).
struct DgradA {
  int n[2], hi[2], wi[2];
  bool rok[2];
  int lsl;
  int co, kw, kh;

  DEVINL void init(const Shape& sh, int m0, long long Mtot) {
    const int t = threadIdx.x;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = p * THREADS + t;
      const int row = idx >> 2;
      const long long m = m0 + row;
      wi[p] = (int)(m % sh.W);
      const long long q = m / sh.W;
      hi[p] = (int)(q % sh.H);
      n[p] = (int)(q / sh.H);
      rok[p] = m < Mtot;
    }
    lsl = (t & 3);
    co = 0; kw = 0; kh = 0;
  }

  DEVINL void stage(bf16* __restrict__ lds, const bf16* __restrict__ dy,
                    const Shape& sh) {
    const int t = threadIdx.x;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = p * THREADS + t;
      const int row = idx >> 2;
      const int sl = lsl ^ ((row >> 2) & 3);
      const int ho = hi[p] + sh.pad - kh;
      const int wo = wi[p] + sh.pad - kw;
      const bool ok = rok[p] && ho >= 0 && ho < sh.HO && wo >= 0 && wo < sh.WO;
      const bf16* src = ok
          ? dy + (((long long)n[p] * sh.HO + ho) * sh.WO + wo) * sh.Cout + co +
                sl * 8
          : (const bf16*)g_zero16;
      glds16(src, lds, p);
    }
    co += KHALF;
    if (co >= sh.Cout) {
      co = 0;
      if (++kw == sh.KW) { kw = 0; ++kh; }
    }
  }
};

// The gemm8 schedule with a templated A stager. B always from a dense
// k-contiguous matrix (weights fwd / w_t dgrad). C written [M, Ntot].
template <typename ASTAGER>
DEVINL void conv8_body(const bf16* __restrict__ A_src,
                       const bf16* __restrict__ B_src, bf16* __restrict__ C,
                       const Shape& sh, long long Mtot, int Ntot, int Ktot,
                       bf16* lds) {
  auto a_slot = [&](int s) -> bf16* { return lds + s * 2 * SLOT; };
  auto b_slot = [&](int s) -> bf16* { return lds + s * 2 * SLOT + SLOT; };

  int tile_n, tile_m;
  {
    const int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    if (sh.swz && nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = bid & 7, idx = bid >> 3;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    tile_n = bid % gridDim.x;
    tile_m = bid / gridDim.x;
  }
  const int m0 = tile_m * TM, n0 = tile_n * TN;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm0 = (wid >> 2) * 128;
  const int wn0 = (wid & 3) * 64;

  f32x4 acc[8][4] = {};

  ASTAGER sa;
  sa.init(sh, m0, Mtot);

  const int nkh = Ktot / KHALF;
  const int pro = nkh < 3 ? nkh : 3;
  // prologue staging: A state advances inside sa.stage; B k-offsets are
  // explicit
  for (int j = 0; j < pro; ++j) {
    sa.stage(a_slot(j & 3), A_src, sh);
    stage_B(b_slot(j & 3), B_src, Ktot, n0, j * KHALF);
  }

  const int arow = lane & 15;
  const int kfrag = (lane >> 4) * 8;

  for (int kh = 0; kh < nkh; ++kh) {
    const int s = kh & 3;
    const bf16* As = a_slot(s);
    const bf16* Bs = b_slot(s);

    if (kh + 3 < nkh) sa.stage(a_slot((kh + 3) & 3), A_src, sh);
    {
      int ahead = 0;
      if (kh + 1 <= nkh - 1) ++ahead;
      if (kh + 2 <= nkh - 1) ++ahead;
      int stages = 2 * ahead + (kh + 3 <= nkh - 1 ? 1 : 0);
      vmwait(2 * stages);
    }
    __builtin_amdgcn_s_barrier();

    s16x8 a_frag[8], b_frag[4];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
      a_frag[mf] = *reinterpret_cast<const s16x8*>(
          &As[soff(wm0 + mf * 16 + arow, kfrag)]);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &Bs[soff(wn0 + nf * 16 + arow, kfrag)]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    if (kh + 3 < nkh)
      stage_B(b_slot((kh + 3) & 3), B_src, Ktot, n0, (kh + 3) * KHALF);
#pragma unroll
    for (int nf = 2; nf < 4; ++nf)
      b_frag[nf] = *reinterpret_cast<const s16x8*>(
          &Bs[soff(wn0 + nf * 16 + arow, kfrag)]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 2; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mf], b_frag[nf], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int col_in_frag = lane & 15;
  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
    const int col = n0 + wn0 + nf * 16 + col_in_frag;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long long row = m0 + wm0 + mf * 16 + row_base + r;
        if (row < Mtot)
          C[row * Ntot + col] = (bf16)acc[mf][nf][r];
      }
  }
}

__global__ __launch_bounds__(THREADS) void conv_fwd_8ph_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    bf16* __restrict__ y, Shape sh) {
  __shared__ bf16 lds[4 * 2 * SLOT];
  const long long Mtot = (long long)sh.N * sh.HO * sh.WO;
  conv8_body<FwdA>(x, w, y, sh, Mtot, sh.Cout, sh.KH * sh.KW * sh.Cin, lds);
}

__global__ __launch_bounds__(THREADS) void conv_dgrad_8ph_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ w_t,
    bf16* __restrict__ dx, Shape sh) {
  __shared__ bf16 lds[4 * 2 * SLOT];
  const long long Mtot = (long long)sh.N * sh.H * sh.W;
  conv8_body<DgradA>(dy, w_t, dx, sh, Mtot, sh.Cin, sh.KH * sh.KW * sh.Cout,
                     lds);
}

}  // namespace c8

#include "launchers.h"

// fwd: y[M=N*HO*WO, Cout]; eligible when Cout % 256 == 0, Cin % 32 == 0,
// K % 64 == 0 (k-halves pair up) and the pixel grid fills the chip.
bool launch_conv_fwd_8ph(const void* x, const void* w, void* y, int N, int H,
                         int W, int Cin, int Cout, int KH, int KW, int stride,
                         int pad, hipStream_t s) {
  const int HO = (H + 2 * pad - KH) / stride + 1;
  const int WO = (W + 2 * pad - KW) / stride + 1;
  const long long M = (long long)N * HO * WO;
  const int Ktot = KH * KW * Cin;
  if (Cout % c8::TN != 0 || Cin % 32 != 0 || Ktot % 64 != 0) return false;
  const long long mt = (M + c8::TM - 1) / c8::TM;
  if (mt * (Cout / c8::TN) < 256) return false;   // 2-phase fills better
  c8::Shape sh{N, H, W, Cin, Cout, KH, KW, stride, pad, HO, WO, 1};
  dim3 grid(Cout / c8::TN, (unsigned)mt);
  hipLaunchKernelGGL(c8::conv_fwd_8ph_kernel, grid, dim3(c8::THREADS), 0, s,
                     (const bf16*)x, (const bf16*)w, (bf16*)y, sh);
  return true;
}

// dgrad: dx[M=N*H*W, Cin]; stride-1 only (strided dgrad keeps the 2-phase
// one-tap stager). w_t is the [Cin, KH*KW*Cout] transposed copy.
bool launch_conv_dgrad_8ph(const void* dy, const void* w_t, void* dx, int N,
                           int H, int W, int Cin, int Cout, int KH, int KW,
                           int stride, int pad, hipStream_t s) {
  if (stride != 1) return false;
  const int HO = (H + 2 * pad - KH) / stride + 1;
  const int WO = (W + 2 * pad - KW) / stride + 1;
  const long long M = (long long)N * H * W;
  const int Ktot = KH * KW * Cout;
  if (Cin % c8::TN != 0 || Cout % 32 != 0 || Ktot % 64 != 0) return false;
  const long long mt = (M + c8::TM - 1) / c8::TM;
  if (mt * (Cin / c8::TN) < 256) return false;
  c8::Shape sh{N, H, W, Cin, Cout, KH, KW, stride, pad, HO, WO, 1};
  dim3 grid(Cin / c8::TN, (unsigned)mt);
  hipLaunchKernelGGL(c8::conv_dgrad_8ph_kernel, grid, dim3(c8::THREADS), 0, s,
                     (const bf16*)dy, (const bf16*)w_t, (bf16*)dx, sh);
  return true;
}
