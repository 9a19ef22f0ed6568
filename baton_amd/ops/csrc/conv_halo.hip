// Shared-halo 3x3 conv (gfx950): fwd + dgrad for the small-channel early
// ResNet layers (stride 1, pad 1).
//
// The implicit-GEMM stagers gather each activation element once PER TAP
// (9 16-B loads per element); at 64-128 channels those layers measure
// ~350-500 TF and are load-ISSUE-bound (~10 B/cyc/CU of gather issue, not
// HBM). Here a block stages its 128-pixel tile ONCE as a padded halo
// image — (R+2) x (W+2) pixel positions x 32 channels, zero pad columns/
// rows written explicitly — and all 9 taps read shifted rows of that one
// image: 1/9th the gather instructions. The weight operand skips LDS
// entirely: B-fragments are contiguous 16-B reads of the (L1/L2-resident,
// tens-of-KB) weight tensor.
//
// Geometry: block = [128 pixels] x [64 filters], 4 waves (each 32 px x
// 64 f), MFMA 16x16x32. Eligible when the 128-pixel tile never crosses an
// image: (H*W) % 128 == 0 (CIFAR 32x32 / 16x16 layers — the bigger-channel
// 8x8 / 4x4 layers run on conv8's 256^2 pipeline instead).
#include "common.h"

namespace ch {

constexpr int THREADS = 256;      // 4 waves
constexpr int PXT = 128;          // pixel tile
constexpr int NT = 64;            // filter tile
constexpr int CK = 32;            // channels per k-step

// halo image: [(R+2)*(W+2) positions][32 ch], 16-B slot XOR by position
DEVINL int hoff(int pos, int ch) {
  return pos * CK + ((((ch >> 3) ^ pos) & 3) << 3) + (ch & 7);
}

struct Shape {
  int N, H, W, Cin, Cout, HR, WR, swz;  // HR = R+2 halo rows, WR = W+2
};

// FLIP=false: forward (A = x); FLIP=true: dgrad (A = dy, tap shift 1-kh).
// B is the weight tensor PRE-PERMUTED into fragment order by the binding:
// wf[tap][kstep][co16-block][4][16][8] — one wave B-fragment load is then
// 64 lanes x 16 B CONTIGUOUS (the naive [Cout][3][3][Cin] layout made
// every load a 16-cacheline gather, which measured slower than the tap
// gathers this kernel replaces). KDIM = A's channel count; NDIM = Y cols.
template <bool FLIP>
__global__ __launch_bounds__(THREADS) void conv_halo_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    bf16* __restrict__ Y, Shape sh, int KDIM, int NDIM) {
  extern __shared__ __attribute__((aligned(16))) bf16 lds[];
  const int hsz = sh.HR * sh.WR;          // halo positions
  bf16* hbuf[2] = {lds, lds + hsz * CK};

  const int tile_px = blockIdx.x;
  const int n0 = blockIdx.y * NT;
  const long long p0 = (long long)tile_px * PXT;
  // tile start decode (tile never crosses an image: (H*W) % 128 == 0)
  const int W = sh.W, H = sh.H;
  const int img = (int)(p0 / ((long long)H * W));
  const int prow0 = (int)((p0 / W) % H);  // first pixel row's ho

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm0 = wid * 32;               // wave's 32 pixels
  const int hi16 = lane >> 4;
  const int l15 = lane & 15;

  // per-lane pixel decode for the two A fragments (k-invariant)
  int apos[2];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
    const int p = wm0 + mf * 16 + l15;    // pixel within tile
    const int r = p / W, c = p % W;
    apos[mf] = (r + 1) * sh.WR + (c + 1); // halo position of the pixel
  }

  const long long img_base = (long long)img * H * W * KDIM;

  f32x4 acc[2][4] = {};

  const int nks = KDIM / CK;
  // ---- halo staging: one 16-B chunk per (position, ch-slot) ----
  auto stage = [&](bf16* hb, int k0) {
    const int chunks = hsz * (CK / 8);
    for (int idx = threadIdx.x; idx < chunks; idx += THREADS) {
      const int slot = idx & 3;
      const int pos = idx >> 2;
      const int hr = pos / sh.WR - 1;     // halo row (-1 .. R)
      const int hc = pos % sh.WR - 1;     // halo col (-1 .. W)
      const int ho = prow0 + hr;
      s16x8 v = s16x8{};
      if (ho >= 0 && ho < H && hc >= 0 && hc < W) {
        v = *reinterpret_cast<const s16x8*>(
            &A[img_base + ((long long)ho * W + hc) * KDIM + k0 + slot * 8]);
      }
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)
              &hb[hoff(pos, slot * 8)]) = v;
    }
  };

  stage(hbuf[0], 0);
  __syncthreads();
  for (int s = 0; s < nks; ++s) {
    const bf16* hb = hbuf[s & 1];
    if (s + 1 < nks) stage(hbuf[(s + 1) & 1], (s + 1) * CK);
    const int kf = hi16 * 8;              // fragment k offset
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      const int kh = t / 3, kw = t % 3;
      const int dh = FLIP ? 1 - kh : kh - 1;
      const int dw = FLIP ? 1 - kw : kw - 1;
      const int shift = dh * sh.WR + dw;
      s16x8 af[2], bf[4];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *reinterpret_cast<const s16x8*>(
            &hb[hoff(apos[mf] + shift, kf)]);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const long long woff =
            (((long long)t * nks + s) * (NDIM >> 4) + (n0 >> 4) + nf) * 512 +
            lane * 8;
        bf[nf] = *reinterpret_cast<const s16x8*>(&B[woff]);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bf[nf], acc[mf][nf], 0, 0, 0);
    }
    __syncthreads();
  }

  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long long px = p0 + wm0 + mf * 16 + row_base + r;
        const int col = n0 + nf * 16 + l15;
        Y[px * NDIM + col] = (bf16)acc[mf][nf][r];
      }
}

}  // namespace ch

#include "launchers.h"

// fwd: Y[N*H*W, Cout] = conv3x3(x, w), stride 1 pad 1.
bool launch_conv_halo_fwd(const void* x, const void* w, void* y, int N, int H,
                          int W, int Cin, int Cout, hipStream_t s) {
  if ((long long)H * W % ch::PXT != 0) return false;
  if (Cin % ch::CK != 0 || Cout % ch::NT != 0) return false;
  const int R = ch::PXT / W;
  if (R * W != ch::PXT || R + 2 > H + 2) return false;
  ch::Shape sh{N, H, W, Cin, Cout, R + 2, W + 2, 1};
  const size_t smem = 2u * (R + 2) * (W + 2) * ch::CK * sizeof(bf16);
  if (smem > 160 * 1024) return false;
  dim3 grid((unsigned)((long long)N * H * W / ch::PXT), Cout / ch::NT);
  hipLaunchKernelGGL((ch::conv_halo_kernel<false>), grid, dim3(ch::THREADS),
                     smem, s, (const bf16*)x, (const bf16*)w, (bf16*)y, sh,
                     Cin, Cout);
  return true;
}

// dgrad: dx[N*H*W, Cin] from dy and w_t [Cin][3][3][Cout].
bool launch_conv_halo_dgrad(const void* dy, const void* w_t, void* dx, int N,
                            int H, int W, int Cin, int Cout, hipStream_t s) {
  if ((long long)H * W % ch::PXT != 0) return false;
  if (Cout % ch::CK != 0 || Cin % ch::NT != 0) return false;
  const int R = ch::PXT / W;
  if (R * W != ch::PXT || R + 2 > H + 2) return false;
  ch::Shape sh{N, H, W, Cin, Cout, R + 2, W + 2, 1};
  const size_t smem = 2u * (R + 2) * (W + 2) * ch::CK * sizeof(bf16);
  if (smem > 160 * 1024) return false;
  dim3 grid((unsigned)((long long)N * H * W / ch::PXT), Cin / ch::NT);
  hipLaunchKernelGGL((ch::conv_halo_kernel<true>), grid, dim3(ch::THREADS),
                     smem, s, (const bf16*)dy, (const bf16*)w_t, (bf16*)dx, sh,
                     Cout, Cin);
  return true;
}
