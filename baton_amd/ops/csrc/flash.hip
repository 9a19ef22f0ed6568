// Fused flash attention (gfx950, bf16): forward + backward.
//
// Replaces the materialized S x S score path (batched GEMMs + standalone
// softmax in attention.hip) for the transformer configs: per Q-tile the
// kernel walks K/V tiles with an online softmax, so scores never touch HBM
// and the softmax/transpose/dP round-trips disappear from the profile
// (reference computation: torch softmax(QK^T)V inside
// nn.MultiheadAttention-equivalents — /root/reference has no attention at
// all, this serves the BERT/Llama BASELINE configs).
//
// Structure (CDNA4 guide §B "fused attention prefill" recipe, adapted):
//   * workgroup = 4 waves x 32 q rows = 128 q rows of one (batch, q-head);
//   * swapped QK^T: S^T = mfma(A=K, B=Q^T) per 32x32x16 — BOTH fragments
//     are contiguous row-major 16-B reads, and the q row is LANE-local
//     (col j = lane&31), so the whole online softmax runs in registers:
//     per-row state (m, l) is one scalar per lane, the row max needs one
//     __shfl_xor(32) to merge the two half-waves;
//   * P -> bf16 B-fragments via float2-to-bf16x2 packs + permlane32_swap
//     (guide T12): the half-waves hold complementary kv quads, one swap
//     per dword pair rebuilds the mfma fragment layout;
//   * PV accumulates O^T = mfma(A=V^T, B=P^T): V is staged TRANSPOSED into
//     LDS ([dh][kv] image) so the A-fragment read is again a contiguous
//     16-B row read; O^T keeps q lane-local, so the online rescale
//     o *= alpha is a lane-local scalar multiply;
//   * K and V^T LDS images are 16-B-slot XOR-swizzled (guide T2 / G4):
//     the fragment reads walk rows per lane at a fixed column slot, which
//     is the 8-16-way-conflict pattern on power-of-2 row strides;
//   * K/V staging is double-buffered with the async-STAGE split (T14):
//     next tile's global loads issue before the compute phase, the LDS
//     write lands after the barrier.
//
// Backward (two-pass recompute, FlashAttention-2 style, no fp32 atomics):
//   * a tiny delta kernel: delta[q] = rowsum(dO o O);
//   * dQ kernel (q-tile-major, same geometry as forward): recomputes
//     P^T = exp(scale*S^T - lse), dP^T = mfma(A=V, B=dO^T-frags),
//     dS^T = P^T o (dP^T - delta) * scale, dQ^T += mfma(A=K^T, B=dS^T);
//   * dK/dV kernel (kv-tile-major, each wave owns 32 kv rows): S/P/dP in
//     the mirrored orientation (kv lane-local), dV^T += mfma(A=dO^T, B=P),
//     dK^T += mfma(A=Q^T, B=dS) with Q/dO staged row-major AND transposed.
#include "common.h"

namespace fa {

constexpr int THREADS = 256;       // 4 waves
constexpr int QW = 32;             // q rows per wave
constexpr int QB = 128;            // q rows per block
constexpr int KVB = 64;            // kv tile

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned int u32x2 __attribute__((ext_vector_type(2)));

DEVINL float bf2f(short h) {
  return __uint_as_float(((unsigned int)(unsigned short)h) << 16);
}

// pack two f32 into one dword of 2 bf16 (compiler emits v_cvt_pk_bf16_f32)
DEVINL unsigned int pk_bf16(float lo, float hi) {
  __hip_bfloat162 t = __float22bfloat162_rn(float2{lo, hi});
  return *reinterpret_cast<unsigned int*>(&t);
}

// K image: [KVB][DH] with 16-B slot XOR (row stride is power of two; the
// fragment read is row-per-lane at a fixed slot -> up to 16-way without it)
template <int DH>
DEVINL int koff(int kv, int d) {
  constexpr int XM = (DH / 8 > 16 ? 16 : DH / 8) - 1;
  return kv * DH + ((((d >> 3) ^ kv) & XM) << 3) + (d & 7);
}
// V^T image: [DH][KVB], same treatment (rows are d, cols are kv)
DEVINL int voff(int d, int kv) {
  return d * KVB + ((((kv >> 3) ^ d) & 7) << 3) + (kv & 7);
}

DEVINL s16x8 lds128(const bf16* p) {
  return *reinterpret_cast<const __attribute__((address_space(3)))
             s16x8*>((const __attribute__((address_space(3))) bf16*)p);
}

// ---- forward ---------------------------------------------------------------
//
// grid: (ceil(S/QB), B*H). q/k/v strided [B,S,(kv)h,DH]; o [B,S,H,DH]
// (d contiguous); lse [B*H, S] fp32 = m + log(l).
template <int DH, bool CAUSAL>
__global__ __launch_bounds__(THREADS) void flash_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE, int S, int H, int G, float scale,
    long long sqB, long long sqS, long long sqH,
    long long skB, long long skS, long long skH,
    long long svB, long long svS, long long svH,
    long long soB, long long soS, long long soH) {
  constexpr int F = DH / 16;        // QK^T mfma count per 32-kv subtile
  constexpr int DB = DH / 32;       // O^T 32-row blocks
  constexpr int NCK = DH / 32;      // 16-B staging chunks per thread/operand

  __shared__ bf16 lds[2 * KVB * DH + 2 * DH * KVB];
  auto kbuf = [&](int b) -> bf16* { return lds + b * KVB * DH; };
  auto vbuf = [&](int b) -> bf16* { return lds + 2 * KVB * DH + b * DH * KVB; };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvh = h / G;

  const bf16* Qp = Q + b * sqB + h * sqH;
  const bf16* Kp = K + b * skB + kvh * skH;
  const bf16* Vp = V + b * svB + kvh * svH;
  bf16* Op = O + b * soB + h * soH;

  const int q0 = qblk * QB;
  const int qg = q0 + wid * QW + lq;          // this lane's q row
  const bool qvalid = qg < S;
  const int qmax_w = min(q0 + wid * QW + QW - 1, S - 1);

  // Q^T B-fragments: lane holds Q[qg][f*16 + hi*8 .. +7]
  s16x8 qf[F];
#pragma unroll
  for (int f = 0; f < F; ++f) {
    if (qvalid)
      qf[f] = *reinterpret_cast<const s16x8*>(
          &Qp[(long long)qg * sqS + f * 16 + hi * 8]);
    else
      qf[f] = s16x8{};
  }

  f32x16 o_acc[DB] = {};
  float m_run = -INFINITY;
  float l_run = 0.f;

  const int nkv_block = CAUSAL ? min(S, q0 + QB) : S;
  const int ntile = (nkv_block + KVB - 1) / KVB;

  // ---- staging helpers (register-staged; T14 split) ----
  s16x8 kreg[NCK], vreg[NCK];
  auto stage_load = [&](int t) {
    const int k0 = t * KVB;
#pragma unroll
    for (int c = 0; c < NCK; ++c) {
      const int idx = c * THREADS + tid;
      const int kv = idx / (DH / 8);
      const int slot = idx % (DH / 8);
      const int kvg = k0 + kv;
      if (kvg < S) {
        kreg[c] = *reinterpret_cast<const s16x8*>(
            &Kp[(long long)kvg * skS + slot * 8]);
        vreg[c] = *reinterpret_cast<const s16x8*>(
            &Vp[(long long)kvg * svS + slot * 8]);
      } else {
        kreg[c] = s16x8{};
        vreg[c] = s16x8{};
      }
    }
  };
  auto stage_write = [&](int buf) {
    bf16* kb = kbuf(buf);
    bf16* vb = vbuf(buf);
#pragma unroll
    for (int c = 0; c < NCK; ++c) {
      const int idx = c * THREADS + tid;
      const int kv = idx / (DH / 8);
      const int slot = idx % (DH / 8);
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)&kb[koff<DH>(kv, slot * 8)]) =
          kreg[c];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vb[voff(slot * 8 + j, kv)] = ((const bf16*)&vreg[c])[j];
    }
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();

  for (int t = 0; t < ntile; ++t) {
    const int cur = t & 1;
    const int k0 = t * KVB;
    if (t + 1 < ntile) stage_load(t + 1);   // issue early (T14)

    const bool active = !CAUSAL || k0 <= qmax_w;
    float p0[16], p1[16];
    if (active) {
      const bf16* kb = kbuf(cur);
      f32x16 sa0 = {}, sa1 = {};
#pragma unroll
      for (int f = 0; f < F; ++f) {
        const s16x8 a0 = lds128(&kb[koff<DH>(lq, f * 16 + hi * 8)]);
        const s16x8 a1 = lds128(&kb[koff<DH>(32 + lq, f * 16 + hi * 8)]);
        sa0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, qf[f], sa0, 0, 0, 0);
        sa1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, qf[f], sa1, 0, 0, 0);
      }
      // mask + scale
      float tm = -INFINITY;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvl = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int kv_g0 = k0 + kvl;
        const int kv_g1 = k0 + 32 + kvl;
        const bool v0 = kv_g0 < S && (!CAUSAL || kv_g0 <= qg);
        const bool v1 = kv_g1 < S && (!CAUSAL || kv_g1 <= qg);
        p0[r] = v0 ? sa0[r] * scale : -1e30f;
        p1[r] = v1 ? sa1[r] * scale : -1e30f;
        tm = fmaxf(tm, fmaxf(p0[r], p1[r]));
      }
      tm = fmaxf(tm, __shfl_xor(tm, 32, 64));
      const float mn = fmaxf(m_run, tm);
      const float alpha = __expf(m_run - mn);
      m_run = mn;
      float rs = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p0[r] = __expf(p0[r] - mn);
        p1[r] = __expf(p1[r] - mn);
        rs += p0[r] + p1[r];
      }
      l_run = l_run * alpha + rs + __shfl_xor(rs, 32, 64);
#pragma unroll
      for (int db = 0; db < DB; ++db)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;

      // P -> bf16 fragments (T12 half-swap); frag[st][halfk]
      s16x8 pf[2][2];
#pragma unroll
      for (int st = 0; st < 2; ++st) {
        const float* p = st == 0 ? p0 : p1;
#pragma unroll
        for (int hk = 0; hk < 2; ++hk) {
          const int o8 = hk * 8;
          unsigned int A = pk_bf16(p[o8 + 0], p[o8 + 1]);
          unsigned int B = pk_bf16(p[o8 + 2], p[o8 + 3]);
          unsigned int C = pk_bf16(p[o8 + 4], p[o8 + 5]);
          unsigned int D = pk_bf16(p[o8 + 6], p[o8 + 7]);
          {
            u32x2 r = __builtin_amdgcn_permlane32_swap(A, C, false, false);
            A = r[0]; C = r[1];
          }
          {
            u32x2 r = __builtin_amdgcn_permlane32_swap(B, D, false, false);
            B = r[0]; D = r[1];
          }
          unsigned int dw[4] = {A, B, C, D};
          pf[st][hk] = *reinterpret_cast<const s16x8*>(dw);
        }
      }
      // PV: O^T += V^T-frag x P-frag
      const bf16* vbp = vbuf(cur);
#pragma unroll
      for (int db = 0; db < DB; ++db) {
#pragma unroll
        for (int st = 0; st < 2; ++st)
#pragma unroll
          for (int hk = 0; hk < 2; ++hk) {
            const s16x8 av = lds128(
                &vbp[voff(db * 32 + lq, st * 32 + hk * 16 + hi * 8)]);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                av, pf[st][hk], o_acc[db], 0, 0, 0);
          }
      }
    }
    if (t + 1 < ntile) {
      __syncthreads();                 // everyone done reading buf cur^1
      stage_write((t + 1) & 1);        // write-late (T14)
      __syncthreads();
    }
  }

  if (!qvalid) return;
  const float inv = l_run > 0.f ? 1.f / l_run : 0.f;
  if (lane < 32) LSE[(long long)bh * S + qg] = m_run + __logf(fmaxf(l_run, 1e-30f));
  bf16* orow = Op + (long long)qg * soS;
#pragma unroll
  for (int db = 0; db < DB; ++db)
#pragma unroll
    for (int qd = 0; qd < 4; ++qd) {
      const int d0 = db * 32 + qd * 8 + hi * 4;
      const unsigned int w0 =
          pk_bf16(o_acc[db][qd * 4 + 0] * inv, o_acc[db][qd * 4 + 1] * inv);
      const unsigned int w1 =
          pk_bf16(o_acc[db][qd * 4 + 2] * inv, o_acc[db][qd * 4 + 3] * inv);
      *reinterpret_cast<u32x2*>(&orow[d0]) = u32x2{w0, w1};
    }
}

// ---- backward --------------------------------------------------------------

// delta[bh, q] = rowsum(dO o O) — one wave per row, vectorized 16-B loads.
// grid: (ceil(S/4), B*H); one wave per (bh, q) row. DH is a multiple of 8,
// so lanes 0..DH/8-1 each reduce one 16-B chunk.
template <int DH>
__global__ __launch_bounds__(256) void flash_delta_kernel(
    const bf16* __restrict__ dO, const bf16* __restrict__ O,
    float* __restrict__ DELTA, int S, int H,
    long long doB, long long doS, long long doH,
    long long oB, long long oS, long long oH) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int q = blockIdx.x * 4 + wid;
  if (q >= S) return;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const bf16* dop = dO + b * doB + h * doH + (long long)q * doS;
  const bf16* op = O + b * oB + h * oH + (long long)q * oS;
  float acc = 0.f;
  if (lane * 8 < DH) {
    float fd[8], fo[8];
    vload16(dop + lane * 8, fd);
    vload16(op + lane * 8, fo);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc = fmaf(fd[j], fo[j], acc);
  }
  acc = wave_reduce_sum(acc);
  acc = __shfl(acc, 0, 64);
  if (lane == 0) DELTA[(long long)bh * S + q] = acc;
}

// dQ kernel: q-tile-major (same block geometry as forward). Per kv tile:
// recompute P^T from lse, dP^T = mfma(V, dO^T), dS^T, then
// dQ^T += mfma(K^T, dS^T-frags). No atomics: dQ accumulates in registers.
template <int DH, bool CAUSAL>
__global__ __launch_bounds__(THREADS) void flash_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dQ, int S, int H, int G, float scale,
    long long sqB, long long sqS, long long sqH,
    long long skB, long long skS, long long skH,
    long long svB, long long svS, long long svH,
    long long doB, long long doS, long long doH,
    long long dqB, long long dqS, long long dqH) {
  constexpr int F = DH / 16;
  constexpr int DB = DH / 32;
  constexpr int NCK = DH / 32;

  // K rows | V rows | K^T — single-buffered (2-phase per tile)
  __shared__ bf16 lds[3 * KVB * DH];
  bf16* kb = lds;
  bf16* vb = lds + KVB * DH;
  bf16* ktb = lds + 2 * KVB * DH;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int kvh = h / G;
  const bf16* Qp = Q + b * sqB + h * sqH;
  const bf16* Kp = K + b * skB + kvh * skH;
  const bf16* Vp = V + b * svB + kvh * svH;
  const bf16* dOp = dO + b * doB + h * doH;
  bf16* dQp = dQ + b * dqB + h * dqH;

  const int q0 = blockIdx.x * QB;
  const int qg = q0 + wid * QW + lq;
  const bool qvalid = qg < S;
  const int qmax_w = min(q0 + wid * QW + QW - 1, S - 1);

  s16x8 qf[F], dof[F];
#pragma unroll
  for (int f = 0; f < F; ++f) {
    if (qvalid) {
      qf[f] = *reinterpret_cast<const s16x8*>(
          &Qp[(long long)qg * sqS + f * 16 + hi * 8]);
      dof[f] = *reinterpret_cast<const s16x8*>(
          &dOp[(long long)qg * doS + f * 16 + hi * 8]);
    } else {
      qf[f] = s16x8{};
      dof[f] = s16x8{};
    }
  }
  const float lse_q = qvalid ? LSE[(long long)bh * S + qg] : 0.f;
  const float delta_q = qvalid ? DELTA[(long long)bh * S + qg] : 0.f;

  f32x16 dq_acc[DB] = {};

  const int nkv_block = CAUSAL ? min(S, q0 + QB) : S;
  const int ntile = (nkv_block + KVB - 1) / KVB;

  for (int t = 0; t < ntile; ++t) {
    const int k0 = t * KVB;
    // ---- stage K, V (row images) + K^T (transposed) — synchronous
    __syncthreads();
#pragma unroll
    for (int c = 0; c < NCK; ++c) {
      const int idx = c * THREADS + tid;
      const int kv = idx / (DH / 8);
      const int slot = idx % (DH / 8);
      const int kvg = k0 + kv;
      s16x8 kr = s16x8{}, vr = s16x8{};
      if (kvg < S) {
        kr = *reinterpret_cast<const s16x8*>(
            &Kp[(long long)kvg * skS + slot * 8]);
        vr = *reinterpret_cast<const s16x8*>(
            &Vp[(long long)kvg * svS + slot * 8]);
      }
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)&kb[koff<DH>(kv, slot * 8)]) = kr;
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)&vb[koff<DH>(kv, slot * 8)]) = vr;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ktb[voff(slot * 8 + j, kv)] = ((const bf16*)&kr)[j];
    }
    __syncthreads();

    if (CAUSAL && k0 > qmax_w) continue;

    // subtiles processed independently (P comes straight from lse)
    s16x8 dsf[2][2];
#pragma unroll
    for (int st = 0; st < 2; ++st) {
      f32x16 sacc = {}, dpacc = {};
#pragma unroll
      for (int f = 0; f < F; ++f) {
        const s16x8 ak = lds128(&kb[koff<DH>(st * 32 + lq, f * 16 + hi * 8)]);
        const s16x8 av = lds128(&vb[koff<DH>(st * 32 + lq, f * 16 + hi * 8)]);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, qf[f], sacc, 0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, dof[f], dpacc, 0, 0, 0);
      }
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv_g = k0 + st * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const bool vld = kv_g < S && (!CAUSAL || kv_g <= qg);
        const float p = vld ? __expf(sacc[r] * scale - lse_q) : 0.f;
        ds[r] = p * (dpacc[r] - delta_q) * scale;
      }
#pragma unroll
      for (int hk = 0; hk < 2; ++hk) {
        const int o8 = hk * 8;
        unsigned int A = pk_bf16(ds[o8 + 0], ds[o8 + 1]);
        unsigned int B = pk_bf16(ds[o8 + 2], ds[o8 + 3]);
        unsigned int C = pk_bf16(ds[o8 + 4], ds[o8 + 5]);
        unsigned int D = pk_bf16(ds[o8 + 6], ds[o8 + 7]);
        {
          u32x2 r = __builtin_amdgcn_permlane32_swap(A, C, false, false);
          A = r[0]; C = r[1];
        }
        {
          u32x2 r = __builtin_amdgcn_permlane32_swap(B, D, false, false);
          B = r[0]; D = r[1];
        }
        unsigned int dw[4] = {A, B, C, D};
        dsf[st][hk] = *reinterpret_cast<const s16x8*>(dw);
      }
    }
    // dQ^T += K^T x dS^T
#pragma unroll
    for (int db = 0; db < DB; ++db)
#pragma unroll
      for (int st = 0; st < 2; ++st)
#pragma unroll
        for (int hk = 0; hk < 2; ++hk) {
          const s16x8 akt = lds128(
              &ktb[voff(db * 32 + lq, st * 32 + hk * 16 + hi * 8)]);
          dq_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              akt, dsf[st][hk], dq_acc[db], 0, 0, 0);
        }
  }

  if (!qvalid) return;
  bf16* dqrow = dQp + (long long)qg * dqS;
#pragma unroll
  for (int db = 0; db < DB; ++db)
#pragma unroll
    for (int qd = 0; qd < 4; ++qd) {
      const int d0 = db * 32 + qd * 8 + hi * 4;
      const unsigned int w0 =
          pk_bf16(dq_acc[db][qd * 4 + 0], dq_acc[db][qd * 4 + 1]);
      const unsigned int w1 =
          pk_bf16(dq_acc[db][qd * 4 + 2], dq_acc[db][qd * 4 + 3]);
      *reinterpret_cast<u32x2*>(&dqrow[d0]) = u32x2{w0, w1};
    }
}

// dK/dV kernel: kv-tile-major; each wave owns 32 kv rows of a 128-kv block
// and loops 32-row q tiles in the mirrored orientation (kv lane-local):
// S = mfma(Q, K), dP = mfma(dO, V) — all row-major reads; P/dS fragments
// via the same half-swap; dV^T += mfma(dO^T, P), dK^T += mfma(Q^T, dS).
// Outputs are PER-Q-HEAD (GQA callers group-sum), so no atomics.
constexpr int KVBB = 128;           // kv rows per block (32 per wave)

template <int DH, bool CAUSAL>
__global__ __launch_bounds__(THREADS) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dK, bf16* __restrict__ dV, int S, int H, int G,
    float scale,
    long long sqB, long long sqS, long long sqH,
    long long skB, long long skS, long long skH,
    long long svB, long long svS, long long svH,
    long long doB, long long doS, long long doH,
    long long dkB, long long dkS, long long dkH,
    long long dvB, long long dvS, long long dvH) {
  constexpr int F = DH / 16;
  constexpr int DB = DH / 32;

  // K rows [128][DH] | V rows [128][DH] | Q [32][DH] | dO [32][DH] |
  // Q^T [DH][32] | dO^T [DH][32]
  __shared__ bf16 lds[2 * KVBB * DH + 2 * 32 * DH + 2 * DH * 32];
  bf16* kb = lds;
  bf16* vb = lds + KVBB * DH;
  bf16* qb = lds + 2 * KVBB * DH;
  bf16* dob = qb + 32 * DH;
  bf16* qtb = dob + 32 * DH;
  bf16* dotb = qtb + DH * 32;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lkv = lane & 31;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int kvh = h / G;
  const bf16* Qp = Q + b * sqB + h * sqH;
  const bf16* Kp = K + b * skB + kvh * skH;
  const bf16* Vp = V + b * svB + kvh * svH;
  const bf16* dOp = dO + b * doB + h * doH;
  bf16* dKp = dK + b * dkB + h * dkH;
  bf16* dVp = dV + b * dvB + h * dvH;

  const int kv0 = blockIdx.x * KVBB;
  const int kv_w = kv0 + wid * 32;            // wave's kv base
  const int kvg = kv_w + lkv;                 // lane's kv row
  const bool kvalid = kvg < S;

  // ---- stage block-persistent K/V row images (swizzled)
  {
    const int nchunk = KVBB * DH / 8;         // 16-B chunks
    for (int idx = tid; idx < nchunk; idx += THREADS) {
      const int kv = idx / (DH / 8);
      const int slot = idx % (DH / 8);
      const int g = kv0 + kv;
      s16x8 kr = s16x8{}, vr = s16x8{};
      if (g < S) {
        kr = *reinterpret_cast<const s16x8*>(
            &Kp[(long long)g * skS + slot * 8]);
        vr = *reinterpret_cast<const s16x8*>(
            &Vp[(long long)g * svS + slot * 8]);
      }
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)&kb[koff<DH>(kv, slot * 8)]) = kr;
      *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
          (__attribute__((address_space(3))) bf16*)&vb[koff<DH>(kv, slot * 8)]) = vr;
    }
  }

  f32x16 dv_acc[DB] = {}, dk_acc[DB] = {};

  const int qt0 = CAUSAL ? (kv0 / 32) : 0;    // first q tile that can see kv0
  const int nqt = (S + 31) / 32;

  for (int qt = qt0; qt < nqt; ++qt) {
    const int q0t = qt * 32;
    // ---- stage Q/dO row images + transposes (32 rows)
    __syncthreads();
    {
      const int nchunk = 32 * DH / 8;
      for (int idx = tid; idx < nchunk; idx += THREADS) {
        const int qr = idx / (DH / 8);
        const int slot = idx % (DH / 8);
        const int g = q0t + qr;
        s16x8 qr16 = s16x8{}, dor16 = s16x8{};
        if (g < S) {
          qr16 = *reinterpret_cast<const s16x8*>(
              &Qp[(long long)g * sqS + slot * 8]);
          dor16 = *reinterpret_cast<const s16x8*>(
              &dOp[(long long)g * doS + slot * 8]);
        }
        *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
            (__attribute__((address_space(3))) bf16*)&qb[koff<DH>(qr, slot * 8)]) = qr16;
        *reinterpret_cast<__attribute__((address_space(3))) s16x8*>(
            (__attribute__((address_space(3))) bf16*)&dob[koff<DH>(qr, slot * 8)]) = dor16;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = slot * 8 + j;
          qtb[d * 32 + ((((qr >> 3) ^ d) & 3) << 3) + (qr & 7)] =
              ((const bf16*)&qr16)[j];
          dotb[d * 32 + ((((qr >> 3) ^ d) & 3) << 3) + (qr & 7)] =
              ((const bf16*)&dor16)[j];
        }
      }
    }
    __syncthreads();

    if (CAUSAL && q0t + 31 < kv_w) continue;  // wave sees nothing here

    // ---- S = Q x K^T and dP = dO x V^T (kv lane-local)
    f32x16 sacc = {}, dpacc = {};
#pragma unroll
    for (int f = 0; f < F; ++f) {
      const s16x8 aq = lds128(&qb[koff<DH>(lkv, f * 16 + hi * 8)]);
      const s16x8 ado = lds128(&dob[koff<DH>(lkv, f * 16 + hi * 8)]);
      const s16x8 bk = lds128(&kb[koff<DH>(wid * 32 + lkv, f * 16 + hi * 8)]);
      const s16x8 bv = lds128(&vb[koff<DH>(wid * 32 + lkv, f * 16 + hi * 8)]);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(aq, bk, sacc, 0, 0, 0);
      dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ado, bv, dpacc, 0, 0, 0);
    }
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int q_g = q0t + (r & 3) + 8 * (r >> 2) + 4 * hi;
      const bool vld = q_g < S && kvalid && (!CAUSAL || kvg <= q_g);
      float lse_r = 0.f, del_r = 0.f;
      if (vld) {
        lse_r = LSE[(long long)bh * S + q_g];
        del_r = DELTA[(long long)bh * S + q_g];
      }
      const float p = vld ? __expf(sacc[r] * scale - lse_r) : 0.f;
      pv[r] = p;
      ds[r] = p * (dpacc[r] - del_r) * scale;
    }
    // fragments over q (half-swap, q consecutive)
    s16x8 pfr[2], dsfr[2];
#pragma unroll
    for (int hk = 0; hk < 2; ++hk) {
      const int o8 = hk * 8;
      unsigned int A = pk_bf16(pv[o8 + 0], pv[o8 + 1]);
      unsigned int B = pk_bf16(pv[o8 + 2], pv[o8 + 3]);
      unsigned int C = pk_bf16(pv[o8 + 4], pv[o8 + 5]);
      unsigned int D = pk_bf16(pv[o8 + 6], pv[o8 + 7]);
      {
        u32x2 r = __builtin_amdgcn_permlane32_swap(A, C, false, false);
        A = r[0]; C = r[1];
      }
      {
        u32x2 r = __builtin_amdgcn_permlane32_swap(B, D, false, false);
        B = r[0]; D = r[1];
      }
      unsigned int dw[4] = {A, B, C, D};
      pfr[hk] = *reinterpret_cast<const s16x8*>(dw);

      unsigned int A2 = pk_bf16(ds[o8 + 0], ds[o8 + 1]);
      unsigned int B2 = pk_bf16(ds[o8 + 2], ds[o8 + 3]);
      unsigned int C2 = pk_bf16(ds[o8 + 4], ds[o8 + 5]);
      unsigned int D2 = pk_bf16(ds[o8 + 6], ds[o8 + 7]);
      {
        u32x2 r = __builtin_amdgcn_permlane32_swap(A2, C2, false, false);
        A2 = r[0]; C2 = r[1];
      }
      {
        u32x2 r = __builtin_amdgcn_permlane32_swap(B2, D2, false, false);
        B2 = r[0]; D2 = r[1];
      }
      unsigned int dw2[4] = {A2, B2, C2, D2};
      dsfr[hk] = *reinterpret_cast<const s16x8*>(dw2);
    }
    // dV^T += dO^T x P ; dK^T += Q^T x dS
#pragma unroll
    for (int db = 0; db < DB; ++db)
#pragma unroll
      for (int hk = 0; hk < 2; ++hk) {
        const int d = db * 32 + lkv;
        const int qcol = hk * 16 + hi * 8;
        const s16x8 adot = lds128(
            &dotb[d * 32 + ((((qcol >> 3) ^ d) & 3) << 3) + (qcol & 7)]);
        const s16x8 aqt = lds128(
            &qtb[d * 32 + ((((qcol >> 3) ^ d) & 3) << 3) + (qcol & 7)]);
        dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            adot, pfr[hk], dv_acc[db], 0, 0, 0);
        dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            aqt, dsfr[hk], dk_acc[db], 0, 0, 0);
      }
  }

  if (!kvalid) return;
  bf16* dkrow = dKp + (long long)kvg * dkS;
  bf16* dvrow = dVp + (long long)kvg * dvS;
#pragma unroll
  for (int db = 0; db < DB; ++db)
#pragma unroll
    for (int qd = 0; qd < 4; ++qd) {
      const int d0 = db * 32 + qd * 8 + hi * 4;
      const unsigned int k0w =
          pk_bf16(dk_acc[db][qd * 4 + 0], dk_acc[db][qd * 4 + 1]);
      const unsigned int k1w =
          pk_bf16(dk_acc[db][qd * 4 + 2], dk_acc[db][qd * 4 + 3]);
      const unsigned int v0w =
          pk_bf16(dv_acc[db][qd * 4 + 0], dv_acc[db][qd * 4 + 1]);
      const unsigned int v1w =
          pk_bf16(dv_acc[db][qd * 4 + 2], dv_acc[db][qd * 4 + 3]);
      *reinterpret_cast<u32x2*>(&dkrow[d0]) = u32x2{k0w, k1w};
      *reinterpret_cast<u32x2*>(&dvrow[d0]) = u32x2{v0w, v1w};
    }
}

}  // namespace fa

#include "launchers.h"

bool launch_flash_fwd(const void* Q, const void* K, const void* V, void* O,
                      float* LSE, int B, int S, int H, int G, int DH,
                      float scale, bool causal,
                      const long long* qs, const long long* ks,
                      const long long* vs, const long long* os,
                      hipStream_t s) {
  if (DH != 32 && DH != 64 && DH != 128) return false;
  dim3 grid((S + fa::QB - 1) / fa::QB, B * H);
#define FA_CALL(DHV, CZ)                                                      \
  hipLaunchKernelGGL((fa::flash_fwd_kernel<DHV, CZ>), grid,                   \
                     dim3(fa::THREADS), 0, s, (const bf16*)Q, (const bf16*)K, \
                     (const bf16*)V, (bf16*)O, LSE, S, H, G, scale,           \
                     qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],                \
                     vs[0], vs[1], vs[2], os[0], os[1], os[2])
  if (DH == 32) { if (causal) FA_CALL(32, true); else FA_CALL(32, false); }
  else if (DH == 64) { if (causal) FA_CALL(64, true); else FA_CALL(64, false); }
  else { if (causal) FA_CALL(128, true); else FA_CALL(128, false); }
#undef FA_CALL
  return true;
}

void launch_flash_delta(const void* dO, const void* O, float* delta,
                        int B, int S, int H, int DH,
                        const long long* dos, const long long* os,
                        hipStream_t s) {
  dim3 grid((S + 3) / 4, B * H);
#define FD_CALL(DHV)                                                         \
  hipLaunchKernelGGL((fa::flash_delta_kernel<DHV>), grid, dim3(256), 0, s,   \
                     (const bf16*)dO, (const bf16*)O, delta, S, H,           \
                     dos[0], dos[1], dos[2], os[0], os[1], os[2])
  if (DH == 32) FD_CALL(32);
  else if (DH == 64) FD_CALL(64);
  else FD_CALL(128);
#undef FD_CALL
}

bool launch_flash_bwd_dq(const void* Q, const void* K, const void* V,
                         const void* dO, const float* LSE, const float* DELTA,
                         void* dQ, int B, int S, int H, int G, int DH,
                         float scale, bool causal,
                         const long long* qs, const long long* ks,
                         const long long* vs, const long long* dos,
                         const long long* dqs, hipStream_t s) {
  if (DH != 32 && DH != 64 && DH != 128) return false;
  dim3 grid((S + fa::QB - 1) / fa::QB, B * H);
#define DQ_CALL(DHV, CZ)                                                      \
  hipLaunchKernelGGL((fa::flash_bwd_dq_kernel<DHV, CZ>), grid,                \
                     dim3(fa::THREADS), 0, s, (const bf16*)Q, (const bf16*)K, \
                     (const bf16*)V, (const bf16*)dO, LSE, DELTA, (bf16*)dQ,  \
                     S, H, G, scale, qs[0], qs[1], qs[2], ks[0], ks[1],       \
                     ks[2], vs[0], vs[1], vs[2], dos[0], dos[1], dos[2],      \
                     dqs[0], dqs[1], dqs[2])
  if (DH == 32) { if (causal) DQ_CALL(32, true); else DQ_CALL(32, false); }
  else if (DH == 64) { if (causal) DQ_CALL(64, true); else DQ_CALL(64, false); }
  else { if (causal) DQ_CALL(128, true); else DQ_CALL(128, false); }
#undef DQ_CALL
  return true;
}

bool launch_flash_bwd_dkv(const void* Q, const void* K, const void* V,
                          const void* dO, const float* LSE, const float* DELTA,
                          void* dK, void* dV, int B, int S, int H, int G,
                          int DH, float scale, bool causal,
                          const long long* qs, const long long* ks,
                          const long long* vs, const long long* dos,
                          const long long* dks, const long long* dvs,
                          hipStream_t s) {
  if (DH != 32 && DH != 64 && DH != 128) return false;
  dim3 grid((S + fa::KVBB - 1) / fa::KVBB, B * H);
#define DKV_CALL(DHV, CZ)                                                     \
  hipLaunchKernelGGL((fa::flash_bwd_dkv_kernel<DHV, CZ>), grid,               \
                     dim3(fa::THREADS), 0, s, (const bf16*)Q, (const bf16*)K, \
                     (const bf16*)V, (const bf16*)dO, LSE, DELTA, (bf16*)dK,  \
                     (bf16*)dV, S, H, G, scale, qs[0], qs[1], qs[2], ks[0],   \
                     ks[1], ks[2], vs[0], vs[1], vs[2], dos[0], dos[1],       \
                     dos[2], dks[0], dks[1], dks[2], dvs[0], dvs[1], dvs[2])
  if (DH == 32) { if (causal) DKV_CALL(32, true); else DKV_CALL(32, false); }
  else if (DH == 64) { if (causal) DKV_CALL(64, true); else DKV_CALL(64, false); }
  else { if (causal) DKV_CALL(128, true); else DKV_CALL(128, false); }
#undef DKV_CALL
  return true;
}
