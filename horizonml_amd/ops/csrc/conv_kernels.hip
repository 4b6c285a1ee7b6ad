// Implicit-GEMM convolution kernels for gfx950 (MI355X, CDNA4).
//
// One MFMA tile template serves three modes:
//   MODE 0  GEMM      C[M,N]  = A[M,Kd] · B[N,Kd]^T            (bf16 in, bf16 out)
//   MODE 1  CONV_FWD  y[m,ko] = Σ_{r,s,c} x[patch] · W[ko,r,s,c]   (NHWC / KRSC)
//   MODE 2  CONV_DGRAD dx[m,c] = Σ_{r,s,ko} dz[mapped] · Wr[r,s,c,ko] (RSCK)
//
// Geometry: 64×64 block tile, BK=32 K-steps, 256 threads = 4 waves in 2×2,
// each wave computes a 32×32 sub-tile as 2×2 mfma_f32_16x16x32_bf16
// fragments.  A/B tiles are LDS-staged with +8-element row padding.
//
// Latency design (these conv shapes are tiny — batch 64 CIFAR gives
// M=64..16k, so kernels are latency-bound, not throughput-bound):
//  * double-buffered LDS with ONE barrier per K-step and 2-deep register
//    prefetch: tile k+2's loads are in flight while tile k computes, so
//    the ~900-cycle HBM latency hides even at 1 block/CU;
//  * split-K: when a long K loop would serialize on an underfilled grid
//    (thresholds measured per direction — see pick_splitk /
//    conv_dgrad_splitk), the K reduction is partitioned over blockIdx.z
//    into per-split f32 slabs summed by the consumer pass (stats_reduce /
//    bn_apply forward, cast[+fused BN reduce] backward) — no atomics.
//
// The non-split forward epilogue also accumulates per-channel BN batch
// statistics (Σy, Σy²) via wave shuffle reduction + one atomicAdd per
// 16-lane group, so a training conv+BN block is 2 kernels total
// (north-star fused conv+BN+ReLU).
#include <cstdlib>

#include "common.h"

struct ConvP {
  int Nb, H, W, C, K;      // batch, input spatial, in/out channels
  int Ho, Wo, R, S;        // output spatial, filter
  int str, pad;
  int M, Kd;               // GEMM rows, reduction size
};

// ------------------------------------------------------------- fast div ----
static void magic_u32(unsigned d, unsigned* m, int* sh) {
  if (d <= 1) { *m = 0; *sh = 0; return; }
  int p = 0;
  while ((1ull << p) < d) p++;
  *m = (unsigned)((((unsigned long long)((1ull << p) - d) << 32) / d) + 1);
  *sh = p;
}

// gather divisors per mode: MODE1 {Ho*Wo, Wo, S*C, C};
// MODE2 {H*W, W, S*K, K} (index 3 also serves b_addr's /K %K)
static MagicP make_magic(const ConvP& p, int mode) {
  MagicP mg{};
  unsigned d[4];
  if (mode == 2) {
    d[0] = (unsigned)(p.H * p.W);
    d[1] = (unsigned)p.W;
    d[2] = (unsigned)(p.S * p.K);
    d[3] = (unsigned)p.K;
  } else {
    d[0] = (unsigned)(p.Ho * p.Wo);
    d[1] = (unsigned)p.Wo;
    d[2] = (unsigned)(p.S * p.C);
    d[3] = (unsigned)p.C;
  }
  for (int i = 0; i < 4; i++) {
    mg.d[i] = d[i] ? d[i] : 1;
    magic_u32(mg.d[i], &mg.m[i], &mg.s[i]);
  }
  return mg;
}

// ---------------------------------------------------------------- staging --
template <int MODE>
DEV int a_addr(const ConvP& p, const MagicP& mg, int m, int k, bool& valid) {
  if (MODE == 0) {
    valid = (m < p.M) & (k < p.Kd);
    return m * p.Kd + k;
  } else if (MODE == 1) {
    // x gather: m -> (n,ho,wo), k -> (r,s,c); index decompositions via
    // magic-number division (the emulated u32 divides dominated the VALU)
    unsigned n = fdiv(m, mg, 0), hw = fmod(m, n, mg, 0);
    unsigned ho = fdiv(hw, mg, 1), wo = fmod(hw, ho, mg, 1);
    unsigned r = fdiv(k, mg, 2), rm = fmod(k, r, mg, 2);
    unsigned sx = fdiv(rm, mg, 3), c = fmod(rm, sx, mg, 3);
    int hi = (int)ho * p.str - p.pad + (int)r;
    int wi = (int)wo * p.str - p.pad + (int)sx;
    valid = (m < p.M) & (k < p.Kd) & (hi >= 0) & (hi < p.H) & (wi >= 0) & (wi < p.W);
    return (((int)n * p.H + hi) * p.W + wi) * p.C + (int)c;
  } else {
    // dz gather (dgrad): m -> (n,hi,wi) over input dims, k -> (r,s,ko)
    unsigned n = fdiv(m, mg, 0), hw = fmod(m, n, mg, 0);
    unsigned hi = fdiv(hw, mg, 1), wi = fmod(hw, hi, mg, 1);
    unsigned r = fdiv(k, mg, 2), rm = fmod(k, r, mg, 2);
    unsigned sx = fdiv(rm, mg, 3), ko = fmod(rm, sx, mg, 3);
    int hs = (int)hi + p.pad - (int)r, ws = (int)wi + p.pad - (int)sx;
    bool ok = (m < p.M) & (k < p.Kd) & (hs >= 0) & (ws >= 0) &&
              (hs % p.str == 0) && (ws % p.str == 0);
    int ho = hs / p.str, wo = ws / p.str;  // str is 1 or 2: cheap
    ok = ok && (ho < p.Ho) && (wo < p.Wo);
    valid = ok;
    return (((int)n * p.Ho + ho) * p.Wo + wo) * p.K + (int)ko;
  }
}

template <int MODE>
DEV int b_addr(const ConvP& p, const MagicP& mg, int n, int k, int Ntot,
               bool& valid) {
  if (MODE == 2) {
    // W_rsck[(r*S+s)*C + c][ko] with k = (r*S+s)*K + ko, n = c
    unsigned rs = fdiv(k, mg, 3), ko = fmod(k, rs, mg, 3);
    valid = (n < Ntot) & (k < p.Kd);
    return ((int)rs * p.C + n) * p.K + (int)ko;
  }
  valid = (n < Ntot) & (k < p.Kd);
  return n * p.Kd + k;
}

template <int MODE, bool VEC>
DEV V8 load8_a(const bf16* __restrict__ src, const ConvP& p,
               const MagicP& mg, int m, int k) {
  V8 v;
  if (VEC) {
    bool ok;
    int a = a_addr<MODE>(p, mg, m, k, ok);
    if (ok) v.u = *(const uint4*)(src + a);
    else v.u = uint4{0, 0, 0, 0};
  } else if (MODE == 1) {
    // scalar x-gather (C not a multiple of 8, e.g. the stem's C=3): one
    // (r,s,c) decomposition per vec8 and incremental carry across the 8
    // elements — the naive per-element a_addr costs ~5 integer divides
    // each and dominated the stem conv
    int n = m / (p.Ho * p.Wo), hw = m % (p.Ho * p.Wo);
    int ho = hw / p.Wo, wo = hw % p.Wo;
    int r = k / (p.S * p.C), rm = k % (p.S * p.C);
    int s = rm / p.C, c = rm % p.C;
    int hi = ho * p.str - p.pad + r, wi = wo * p.str - p.pad + s;
    const bool mok = m < p.M;
    int kk = k;
#pragma unroll
    for (int e = 0; e < 8; e++) {
      bool ok = mok && kk < p.Kd && hi >= 0 && hi < p.H && wi >= 0 &&
                wi < p.W;
      v.e[e] = ok ? src[((n * p.H + hi) * p.W + wi) * p.C + c] : (bf16)0.f;
      kk++;
      if (++c == p.C) {
        c = 0;
        ++wi;
        if (++s == p.S) {
          s = 0;
          wi -= p.S;
          ++r;
          ++hi;
        }
      }
    }
  } else {
    for (int e = 0; e < 8; e++) {
      bool ok;
      int a = a_addr<MODE>(p, mg, m, k + e, ok);
      v.e[e] = ok ? src[a] : (bf16)0.f;
    }
  }
  return v;
}

template <int MODE, bool VEC>
DEV V8 load8_b(const bf16* __restrict__ src, const ConvP& p,
               const MagicP& mg, int n, int k, int Ntot) {
  V8 v;
  if (VEC) {
    bool ok;
    int a = b_addr<MODE>(p, mg, n, k, Ntot, ok);
    if (ok) v.u = *(const uint4*)(src + a);
    else v.u = uint4{0, 0, 0, 0};
  } else {
    for (int e = 0; e < 8; e++) {
      bool ok;
      int a = b_addr<MODE>(p, mg, n, k + e, Ntot, ok);
      v.e[e] = ok ? src[a] : (bf16)0.f;
    }
  }
  return v;
}

// ------------------------------------------------------------- main tile --
#define LDA 40  // 32 + 8 pad; row stride 80 B

// SPLIT: each blockIdx.z writes its f32 partial into its own slab
// ws_out[z][M][Ntot] (no atomics, no pre-zeroing — the consumer pass sums
// the slabs); otherwise write bf16 into Y (+ optional stats).
//
// TM×TN block tile (waves in a fixed 2×2 grid, wave tile (TM/2)×(TN/2)):
//  * 64×64 (default) — the LATENCY tile for the CIFAR-shape grids: small
//    LDS footprint, shortest prologue, best when the grid underfills the
//    256 CUs and every block must hide HBM latency alone;
//  * 128×64 / 128×128 — THROUGHPUT tiles for ImageNet-shaped work
//    (selected by pick_tile when the tiled grid still fills the chip):
//    each wave runs 8/16 MFMAs per K-step against the same 6/8 LDS
//    fragment reads, lifting the MFMA:issue ratio that caps the 64×64
//    tile at ~2% of the bf16 peak on ResNet50@224 (r50_224_pmc_mfma.txt).
template <int MODE, bool VECA, bool VECB, bool STATS, bool SPLIT,
          int TM = 64, int TN = 64>
__global__ __launch_bounds__(256) void k_conv_mfma(
    const bf16* __restrict__ A, const bf16* __restrict__ Bw,
    bf16* __restrict__ Y, float* __restrict__ ws_out,
    float* __restrict__ stats, ConvP p, MagicP mg, int Ntot, int kchunk,
    int accum) {
  // Double-buffered LDS, ONE barrier per K-step, 2-deep register
  // prefetch: tile k+2's global loads are in flight while tile k computes,
  // so the ~900-cycle HBM latency hides even at 1 block/CU.
  constexpr int AC = TM / 64;   // A stage chunks per thread (vec8 each)
  constexpr int BC = TN / 64;
  constexpr int MI = TM / 32;   // 16-row m fragments per wave
  constexpr int NI = TN / 32;
  constexpr int WM = TM / 2, WN = TN / 2;
  __shared__ bf16 As[2][TM * LDA];
  __shared__ bf16 Bs[2][TN * LDA];

  const int m0 = blockIdx.y * TM, n0 = blockIdx.x * TN;
  const int kbeg = SPLIT ? blockIdx.z * kchunk : 0;
  const int kend = SPLIT ? min(p.Kd, kbeg + kchunk) : p.Kd;
  const int tid = threadIdx.x;
  const int srow = tid >> 2, scol = (tid & 3) * 8;  // 1 vec8 per thread
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int fr = lane & 15, fk = lane >> 4;

  f32x4 acc[MI][NI] = {};

  // prologue: tile 0 -> LDS[0]; tile 1 -> registers
#pragma unroll
  for (int j = 0; j < AC; j++) {
    V8 a0 = load8_a<MODE, VECA>(A, p, mg, m0 + j * 64 + srow, kbeg + scol);
    *(V8*)&As[0][(j * 64 + srow) * LDA + scol] = a0;
  }
#pragma unroll
  for (int j = 0; j < BC; j++) {
    V8 b0 = load8_b<MODE, VECB>(Bw, p, mg, n0 + j * 64 + srow, kbeg + scol,
                                Ntot);
    *(V8*)&Bs[0][(j * 64 + srow) * LDA + scol] = b0;
  }
  V8 a_nx[AC], b_nx[BC];
  if (kbeg + 32 < kend) {
#pragma unroll
    for (int j = 0; j < AC; j++)
      a_nx[j] = load8_a<MODE, VECA>(A, p, mg, m0 + j * 64 + srow,
                                    kbeg + 32 + scol);
#pragma unroll
    for (int j = 0; j < BC; j++)
      b_nx[j] = load8_b<MODE, VECB>(Bw, p, mg, n0 + j * 64 + srow,
                                    kbeg + 32 + scol, Ntot);
  }
  __syncthreads();

  int buf = 0;
  for (int k0 = kbeg; k0 < kend; k0 += 32) {
    // stage tile k+1 from registers into the OTHER buffer (its readers
    // synchronized at the previous barrier), then issue tile k+2's loads
    if (k0 + 32 < kend) {
#pragma unroll
      for (int j = 0; j < AC; j++)
        *(V8*)&As[buf ^ 1][(j * 64 + srow) * LDA + scol] = a_nx[j];
#pragma unroll
      for (int j = 0; j < BC; j++)
        *(V8*)&Bs[buf ^ 1][(j * 64 + srow) * LDA + scol] = b_nx[j];
      if (k0 + 64 < kend) {
#pragma unroll
        for (int j = 0; j < AC; j++)
          a_nx[j] = load8_a<MODE, VECA>(A, p, mg, m0 + j * 64 + srow,
                                        k0 + 64 + scol);
#pragma unroll
        for (int j = 0; j < BC; j++)
          b_nx[j] = load8_b<MODE, VECB>(Bw, p, mg, n0 + j * 64 + srow,
                                        k0 + 64 + scol, Ntot);
      }
    }
    bf16x8 af[MI], bf[NI];
#pragma unroll
    for (int mi = 0; mi < MI; mi++)
      af[mi] =
          *(const bf16x8*)&As[buf][(wr * WM + mi * 16 + fr) * LDA + fk * 8];
#pragma unroll
    for (int ni = 0; ni < NI; ni++)
      bf[ni] =
          *(const bf16x8*)&Bs[buf][(wc * WN + ni * 16 + fr) * LDA + fk * 8];
#pragma unroll
    for (int mi = 0; mi < MI; mi++)
#pragma unroll
      for (int ni = 0; ni < NI; ni++)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    buf ^= 1;
    __syncthreads();
  }

  // Epilogue. D fragment: col = lane&15 (=fr), row = fk*4 + q.
  float ssum[NI], ssq[NI];
#pragma unroll
  for (int ni = 0; ni < NI; ni++) ssum[ni] = ssq[ni] = 0.f;
#pragma unroll
  for (int mi = 0; mi < MI; mi++) {
#pragma unroll
    for (int ni = 0; ni < NI; ni++) {
      int gn = n0 + wc * WN + ni * 16 + fr;
#pragma unroll
      for (int q = 0; q < 4; q++) {
        int gm = m0 + wr * WM + mi * 16 + fk * 4 + q;
        if (gm < p.M && gn < Ntot) {
          float v = acc[mi][ni][q];
          if (SPLIT) {
            ws_out[(long)blockIdx.z * p.M * Ntot + (long)gm * Ntot + gn] = v;
          } else {
            long yi = (long)gm * Ntot + gn;
            if (accum) v += b2f(Y[yi]);  // fused residual-junction add
            Y[yi] = f2b(v);
            if (STATS) {
              ssum[ni] += v;
              ssq[ni] += v * v;
            }
          }
        }
      }
    }
  }
  if (STATS && !SPLIT && stats != nullptr) {
#pragma unroll
    for (int ni = 0; ni < NI; ni++) {
      float s = ssum[ni] + __shfl_xor(ssum[ni], 16, 64);
      s += __shfl_xor(s, 32, 64);
      float s2 = ssq[ni] + __shfl_xor(ssq[ni], 16, 64);
      s2 += __shfl_xor(s2, 32, 64);
      int gn = n0 + wc * WN + ni * 16 + fr;
      if (fk == 0 && gn < Ntot) {
        atomicAdd(&stats[gn], s);
        atomicAdd(&stats[Ntot + gn], s2);
      }
    }
  }
}

// ------------------------------------------------------------------ wgrad --
// dW[ko][k3] += Σ_m patch(m,k3) · dz(m,ko);  dW is KRSC flat [Ntot][Kd] f32.
// MFMA formulation: the reduction axis is m, so both operands are staged
// TRANSPOSED into LDS ([k3][m] and [ko][m]) during the gather (coalesced
// global vec8 reads, 8 strided b16 LDS writes per thread), which makes the
// MFMA fragment reads contiguous ds_read_b128.  Output tile 64(k3)×64(ko),
// m reduced 32-deep per MFMA step, msplit-way M parallelism.
//
// Epilogue is ATOMIC-FREE (measured: the former per-element f32 atomicAdd
// into dW cost ~12.5 µs per million atomics and dominated wgrad — see
// profiles/bench_r01_opt1_graph_stats.txt): the accumulator tile is
// transposed through LDS into [ko][k3] row order and written with plain
// coalesced float4 accesses — read-modify-write into the pre-zeroed
// gradient when msplit == 1 (single writer per element; += preserves
// multi-backward accumulation), else pure stores into a per-split slab
// ws[z][Ntot][Kd] that the reduce kernel sums into the gradient.
//
// Deferred/batched mode (the flat fast path): all convs' wgrads of one
// backward run as ONE kernel — every task's (tile, z) blocks are in flight
// together, so the chip fills without oversplitting M, and 2 launches
// replace ~40 (bind.cpp::flush_wgrad).
#define LDW 40  // 32 m + 8 pad

// TK3: k3-tile width (64 default; 128 for big-Kd large-M convs — halves
// the Dz re-read traffic and doubles MFMA per staging write).  The out
// image keeps an odd f32 row stride (65 / 131) for conflict-free stores.
template <int TK3, int TKO = 64>
union WgradSmemT {
  struct {
    bf16 At[TK3 * LDW];  // [k3][m]
    bf16 Dt[TKO * LDW];  // [ko][m]
  } s;
  float out[64][TK3 == 64 ? 65 : 131];  // transposed epilogue staging
};
using WgradSmem = WgradSmemT<64, 64>;

template <bool VECA, int TK3 = 64, int TKO = 64>
DEV void wgrad_tile(const bf16* __restrict__ X, const bf16* __restrict__ Dz,
                    float* __restrict__ out, const ConvP& p,
                    const MagicP& mg, int Ntot, int mchunk, int tx, int ty,
                    int z, bool split, WgradSmemT<TK3, TKO>& smem) {
  constexpr int AC = TK3 / 64;   // A staging chunks per thread
  constexpr int DC = TKO / 64;   // Dz staging chunks per thread
  constexpr int MI = TK3 / 32;   // k3 fragments per wave
  constexpr int NI = TKO / 32;   // ko fragments per wave
  bf16* At = smem.s.At;   // [k3][m]
  bf16* Dt = smem.s.Dt;   // [ko][m]
  const int k3_0 = tx * TK3, n0 = ty * TKO;
  const int mbeg = z * mchunk;
  const int mend = min(p.M, mbeg + mchunk);
  const int tid = threadIdx.x;
  const int sm = tid >> 3, sv = (tid & 7) * 8;  // m-lane, 8-wide k3/ko chunk
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int fr = lane & 15, fk = lane >> 4;

  f32x4 acc[MI][NI] = {};

  V8 a_nx[AC], d_nx[DC];
#pragma unroll
  for (int j = 0; j < AC; j++)
    a_nx[j] = load8_a<1, VECA>(X, p, mg, mbeg + sm, k3_0 + j * 64 + sv);
#pragma unroll
  for (int j = 0; j < DC; j++) {
    int m = mbeg + sm, n = n0 + j * 64 + sv;
    if (m < p.M && n < Ntot)
      d_nx[j].u = *(const uint4*)(Dz + (long)m * Ntot + n);
    else
      d_nx[j].u = uint4{0, 0, 0, 0};
  }
  const int smlo = sm & 7, smhi = sm >> 3;
  for (int m0 = mbeg; m0 < mend; m0 += 32) {
    // transposed scatter into LDS: element (m=sm, k3=sv+e) -> At[sv+e][sm']
    // with the m 8-chunk XOR-swizzled by row&3 (8-way -> 2-way write banks)
#pragma unroll
    for (int j = 0; j < AC; j++)
#pragma unroll
      for (int e = 0; e < 8; e++) {
        int row = j * 64 + sv + e;
        int smx = smlo | (((smhi ^ row) & 3) << 3);
        At[row * LDW + smx] = a_nx[j].e[e];
      }
#pragma unroll
    for (int j = 0; j < DC; j++)
#pragma unroll
      for (int e = 0; e < 8; e++) {
        int row = j * 64 + sv + e;
        int smx = smlo | (((smhi ^ row) & 3) << 3);
        Dt[row * LDW + smx] = d_nx[j].e[e];
      }
    __syncthreads();
    if (m0 + 32 < mend) {
#pragma unroll
      for (int j = 0; j < AC; j++)
        a_nx[j] = load8_a<1, VECA>(X, p, mg, m0 + 32 + sm,
                                   k3_0 + j * 64 + sv);
#pragma unroll
      for (int j = 0; j < DC; j++) {
        int m = m0 + 32 + sm, n = n0 + j * 64 + sv;
        if (m < p.M && n < Ntot)
          d_nx[j].u = *(const uint4*)(Dz + (long)m * Ntot + n);
        else
          d_nx[j].u = uint4{0, 0, 0, 0};
      }
    }
    bf16x8 af[MI], bf[NI];
#pragma unroll
    for (int mi = 0; mi < MI; mi++) {
      int row = wr * (TK3 / 2) + mi * 16 + fr;
      af[mi] = *(const bf16x8*)&At[row * LDW + ((fk ^ row) & 3) * 8];
    }
#pragma unroll
    for (int ni = 0; ni < NI; ni++) {
      int row = wc * (TKO / 2) + ni * 16 + fr;
      bf[ni] = *(const bf16x8*)&Dt[row * LDW + ((fk ^ row) & 3) * 8];
    }
#pragma unroll
    for (int mi = 0; mi < MI; mi++)
#pragma unroll
      for (int ni = 0; ni < NI; ni++)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
  }
  // D: col (=ko) = fr, row (=k3) = fk*4+q.  Transpose through the 64-ko
  // LDS image, one pass per 64-ko slab (TKO=128: pass p uses only the
  // wc==p waves' accumulators; every thread joins the global store).
  float* dst = out + (split ? (long)z * Ntot * (long)p.Kd : 0L);
  constexpr int CCH = TK3 / 4;  // k3 span per thread in the store pass
  const int ko_r = tid >> 2, cch = (tid & 3) * CCH;
#pragma unroll
  for (int pass = 0; pass < TKO / 64; pass++) {
    __syncthreads();  // LDS union / previous pass done
    if (TKO == 64 || wc == pass) {
#pragma unroll
      for (int mi = 0; mi < MI; mi++)
#pragma unroll
        for (int ni = 0; ni < NI; ni++) {
          int ko_l = wc * (TKO / 2) + ni * 16 + fr - pass * 64;
#pragma unroll
          for (int q = 0; q < 4; q++)
            smem.out[ko_l][wr * (TK3 / 2) + mi * 16 + fk * 4 + q] =
                acc[mi][ni][q];
        }
    }
    __syncthreads();
    const int gko = n0 + pass * 64 + ko_r;
    if (gko < Ntot) {
      long base = (long)gko * p.Kd + k3_0 + cch;
      if (k3_0 + cch + CCH <= p.Kd) {
#pragma unroll
        for (int e = 0; e < CCH; e += 4) {
          float4 v = *(const float4*)&smem.out[ko_r][cch + e];
          if (!split) {  // accumulate into .grad (single writer; RMW safe)
            float4 d = *(const float4*)&dst[base + e];
            v.x += d.x; v.y += d.y; v.z += d.z; v.w += d.w;
          }
          *(float4*)&dst[base + e] = v;
        }
      } else {
        for (int e = 0; e < CCH && k3_0 + cch + e < p.Kd; e++)
          dst[base + e] =
              smem.out[ko_r][cch + e] + (split ? 0.f : dst[base + e]);
      }
    }
  }
}

template <bool VECA>
__global__ __launch_bounds__(256) void k_wgrad(
    const bf16* __restrict__ X, const bf16* __restrict__ Dz,
    float* __restrict__ dW, ConvP p, MagicP mg, int Ntot, int mchunk) {
  __shared__ WgradSmem smem;
  wgrad_tile<VECA, 64, 64>(X, Dz, dW, p, mg, Ntot, mchunk, blockIdx.x,
                           blockIdx.y, blockIdx.z, gridDim.z > 1, smem);
}

// Σ over msplit wgrad slabs [z][n] -> dW[n] (+=; dW pre-zeroed or fresh).
__global__ __launch_bounds__(256) void k_wgrad_reduce(
    const float* __restrict__ ws, float* __restrict__ dW, long n,
    int msplit) {
  long i4 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (; i4 + 4 <= n; i4 += stride) {
    float4 v = *(const float4*)&ws[i4];
    for (int z = 1; z < msplit; z++) {
      float4 w = *(const float4*)&ws[(long)z * n + i4];
      v.x += w.x; v.y += w.y; v.z += w.z; v.w += w.w;
    }
    float4 d = *(float4*)&dW[i4];
    d.x += v.x; d.y += v.y; d.z += v.z; d.w += v.w;
    *(float4*)&dW[i4] = d;
  }
  // tail (Kd*K not divisible by 4)
  if (blockIdx.x == 0 && threadIdx.x < 4) {
    long tail = n & ~3L;
    for (long i = tail + threadIdx.x; i < n; i += 4) {
      float v = ws[i];
      for (int z = 1; z < msplit; z++) v += ws[(long)z * n + i];
      dW[i] += v;
    }
  }
}

// ------------------------------------------------- batched (deferred) wgrad --
// One launch computes every pending conv's weight gradient: all tasks'
// (tile, z) blocks are in flight together so msplit can stay small
// (mchunk ≈ 512 rows) and the grid still fills 256 CUs without atomics.
// Task tables travel as kernel arguments by value — no staging buffers,
// hipGraph-capture-safe.
#define WG_MAX_TASKS 20

struct WgradTask {
  const bf16* X;
  const bf16* Dz;
  float* out;      // dW when msplit==1, else slab base [msplit][Ntot][Kd]
  ConvP p;
  MagicP mg;
  int Ntot, mchunk, msplit, tx, ty;  // tile counts
  int base;        // first block id of this task
  int vec;
};

struct WgradBatchArgs {
  int n;
  WgradTask t[WG_MAX_TASKS];
};

template <int TK3, int TKO>
__global__ __launch_bounds__(256) void k_wgrad_batched(WgradBatchArgs a) {
  int bid = blockIdx.x;
  int i = 0;
  while (i + 1 < a.n && bid >= a.t[i + 1].base) i++;
  const WgradTask& t = a.t[i];
  int local = bid - t.base;
  int nt = t.tx * t.ty;
  int z = local / nt, rem = local - z * nt;
  int ty = rem / t.tx, tx = rem - ty * t.tx;
  __shared__ WgradSmemT<TK3, TKO> smem;
  if (t.vec)
    wgrad_tile<true, TK3, TKO>(t.X, t.Dz, t.out, t.p, t.mg, t.Ntot,
                               t.mchunk, tx, ty, z, t.msplit > 1, smem);
  else
    wgrad_tile<false, TK3, TKO>(t.X, t.Dz, t.out, t.p, t.mg, t.Ntot,
                                t.mchunk, tx, ty, z, t.msplit > 1, smem);
}

struct WredTask {
  const float* ws;
  float* dW;
  long n;
  int msplit;
  int base;  // first block id; each block covers 1024 floats
};

struct WredBatchArgs {
  int n;
  WredTask t[WG_MAX_TASKS];
};

__global__ __launch_bounds__(256) void k_wgrad_reduce_batched(
    WredBatchArgs a) {
  int bid = blockIdx.x;
  int i = 0;
  while (i + 1 < a.n && bid >= a.t[i + 1].base) i++;
  const WredTask& t = a.t[i];
  long i4 = ((long)(bid - t.base) * 256 + threadIdx.x) * 4;
  if (i4 + 4 <= t.n) {
    float4 v = *(const float4*)&t.ws[i4];
    for (int z = 1; z < t.msplit; z++) {
      float4 w = *(const float4*)&t.ws[(long)z * t.n + i4];
      v.x += w.x; v.y += w.y; v.z += w.z; v.w += w.w;
    }
    float4 d = *(float4*)&t.dW[i4];
    d.x += v.x; d.y += v.y; d.z += v.z; d.w += v.w;
    *(float4*)&t.dW[i4] = d;
  } else {
    for (long j = i4; j < t.n; j++) {
      float v = t.ws[j];
      for (int z = 1; z < t.msplit; z++) v += t.ws[(long)z * t.n + j];
      t.dW[j] += v;
    }
  }
}

// ------------------------------------------------------------- launchers --
static inline int cdiv_h(int a, int b) { return (a + b - 1) / b; }

// Split-K heuristic.  Split-K costs a whole consumer chain (f32 slabs +
// stats_reduce/cast + slab re-reads), so it only pays when a non-split
// launch would serialize a long K loop on an underfilled grid: measured on
// MI355X, non-split wins for Kd ≤ ~1152 even at 32–64 workgroups (the
// K-loop is ~18–36 latency-hidden iterations), while layer3/4 shapes
// (Kd ≥ 2304, ≤16 tiles) still want the split.
static inline int pick_splitk(int tiles, int Kd) {
  // tunables (read once): grid-fill target and minimum Kd that justifies
  // the split-K consumer chain
  static int tile_min = [] {
    const char* e = getenv("HZ_SK_TILES");
    // r02 re-sweep: 256 splits the layer3/4-class long-K forwards too:
    // r50@224 10.01 -> 9.05 ms same-box, CIFAR bs=1024 +3%, bs=64 flat
    return e ? atoi(e) : 256;
  }();
  static int kd_min = [] {
    const char* e = getenv("HZ_SK_KD");
    return e ? atoi(e) : 1024;
  }();
  if (tiles >= tile_min || Kd < kd_min) return 1;
  int sk = cdiv_h(256, tiles);
  int maxsk = cdiv_h(Kd, 32);
  if (sk > maxsk) sk = maxsk;
  if (sk > 32) sk = 32;
  return sk;
}

static inline int pick_tile(int M, int N);
static inline int pick_tile_dgrad(int M, int N);

// Fill-split (HZ_SK_FILL / HZ_SK_FILL_DG, default 0 = off): when the
// throughput 128x128 tile is picked but its grid lands under `fillmin`
// blocks (r50@224 bs=32 mid-layers: 98-196 blocks on 256 CUs), split K
// across blockIdx.z to recover occupancy instead of falling back to the
// latency tile.  The split path pays the f32 workspace + consumer pass,
// so this only makes sense for long-K shapes (kd floor re-used).
static int sk_fill(int dgrad) {
  static int vf = [] {
    const char* e = getenv("HZ_SK_FILL");
    return e ? atoi(e) : 0;
  }();
  static int vd = [] {
    const char* e = getenv("HZ_SK_FILL_DG");
    // swept on r50@224 bs=32 (same box): 8.536 -> 8.485 ms; the fwd
    // variant (HZ_SK_FILL) measured flat and stays off — its split
    // drags the stats pass + slab re-reads
    return e ? atoi(e) : 256;
  }();
  return dgrad ? vd : vf;
}

static int fill_splitk(int blocks128, int Kd, int fillmin) {
  int sk = cdiv_h(fillmin, blocks128);
  int maxsk = cdiv_h(Kd, 32);
  if (sk > maxsk) sk = maxsk;
  if (sk > 8) sk = 8;
  return sk;
}

extern "C" int conv_fwd_splitk(ConvP p) {
  int tiles = cdiv_h(p.K, 64) * cdiv_h(p.M, 64);
  int sk = pick_splitk(tiles, p.Kd);
  if (sk == 1 && sk_fill(0) > 0 && p.Kd >= 512
      && pick_tile(p.M, p.K) == 2) {
    int blocks = cdiv_h(p.M, 128) * cdiv_h(p.K, 128);
    if (blocks < sk_fill(0)) return fill_splitk(blocks, p.Kd, sk_fill(0));
  }
  return sk;
}

extern "C" int conv_dgrad_splitk(ConvP p) {
  // dgrad geometry: M=Nb*H*W rows, Kd=R*S*K, N=C.  dgrad has its own Kd
  // threshold: its split path costs only the cast pass (often fused with
  // the upstream conv's BN reduce), so splitting pays at smaller Kd than
  // the forward's split (which drags a stats pass + slab re-reads).
  static int kd_min_dg = [] {
    const char* e = getenv("HZ_SK_KD_DG");
    return e ? atoi(e) : 512;
  }();
  static int tile_min_dg = [] {
    const char* e = getenv("HZ_SK_TILES_DG");
    return e ? atoi(e) : 256;  // swept: r50@224 9.05 -> 8.89 ms; CIFAR flat
  }();
  int tiles = cdiv_h(p.C, 64) * cdiv_h(p.Nb * p.H * p.W, 64);
  int Kd = p.R * p.S * p.K;
  if (tiles >= tile_min_dg || Kd < kd_min_dg) {
    int M = p.Nb * p.H * p.W;
    if (sk_fill(1) > 0 && Kd >= 512 && pick_tile_dgrad(M, p.C) == 2) {
      int blocks = cdiv_h(M, 128) * cdiv_h(p.C, 128);
      if (blocks < sk_fill(1)) return fill_splitk(blocks, Kd, sk_fill(1));
    }
    return 1;
  }
  int sk = cdiv_h(256, tiles);
  int maxsk = cdiv_h(Kd, 32);
  if (sk > maxsk) sk = maxsk;
  if (sk > 32) sk = 32;
  return sk;
}

// ------------------------------------------------------ stem direct conv --
// Specialized forward for the ImageNet stem (C=3, 7x7, stride 2, pad 3,
// K<=64): the generic implicit-GEMM path must SCALAR-gather x (C=3 defeats
// vec8) and runs short K loops per 64x64 tile — measured 209 us at
// ResNet50@224 bs=32 / ~24 us at the CIFAR parity point.  Here the whole
// weight tensor (<=64x147 bf16) lives in LDS zero-PADDED to 32 k-slots per
// filter row (one K-step == one filter row r), and the needed x rows are
// staged once per block with a 9-element halo — the im2col "gather"
// becomes one linear LDS read, because for fixed r the (s,c) flattening is
// stride-1 in NHWC memory: patch element j = s*3+c sits at
// x[row][(wo*2-3)*3 + j].  Pad-k products are exact zeros (weight slots
// zeroed; the A slack reads stay inside the staged buffer, so finite).
//
// Block: rows_pb output rows x wo_pad columns (<=128 outputs) of ONE
// image, all K channels.  Waves 2x2: wave tile 64(m) x 32(K): 8 MFMA per
// K-step, 7 K-steps.  A-fragment LDS reads are 4x b32 (patch base is only
// 4-byte aligned: wo*6 bf16); W rows use a 232-element pitch (16B-aligned
// rows, 116-dword stride => conflict-free 16-lane b128 reads).
#define STEM_WPITCH 232
template <bool STATS>
__global__ __launch_bounds__(256) void k_stem_conv(
    const bf16* __restrict__ X, const bf16* __restrict__ Wk,
    bf16* __restrict__ Y, float* __restrict__ stats, ConvP p, int rows_pb,
    int wo_pad, int xpitch, int bpi) {
  extern __shared__ bf16 sm[];
  bf16* Wl = sm;                       // [64][STEM_WPITCH]
  bf16* Xl = sm + 64 * STEM_WPITCH;    // [rows_pb*2+5][xpitch]
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int fr = lane & 15, fk = lane >> 4;

  // --- stage weights: W flat [K][147] -> Wl[K][r*32 + j] (j<21), 0 pad --
  for (int i = tid; i < 64 * STEM_WPITCH; i += 256) {
    int ch = i / STEM_WPITCH, col = i - ch * STEM_WPITCH;
    int r = col >> 5, j = col & 31;
    bf16 v = (bf16)0.f;
    if (ch < p.K && r < 7 && j < 21 && col < 7 * 32)
      v = Wk[ch * 147 + r * 21 + j];
    Wl[i] = v;
  }

  // --- stage x rows with 9-element left halo, zero outside [0,W) -------
  const int n = blockIdx.x / bpi;
  const int ho0 = (blockIdx.x - n * bpi) * rows_pb;
  const int nrows = rows_pb * 2 + 5;   // input rows hi0 .. hi0+nrows-1
  const int hi0 = ho0 * 2 - 3;
  const int rowlen = p.W * 3;
  for (int i = tid; i < nrows * xpitch; i += 256) {
    int rr = i / xpitch, col = i - rr * xpitch;
    int hi = hi0 + rr;
    bf16 v = (bf16)0.f;
    int e = col - 9;  // halo: col 9 == x element 0 of the row
    if (hi >= 0 && hi < p.H && e >= 0 && e < rowlen)
      v = X[((long)(n * p.H + hi) * p.W) * 3 + e];
    Xl[i] = v;
  }
  __syncthreads();

  // --- MFMA main loop: 7 K-steps (one filter row each) ------------------
  f32x4 acc[4][2] = {};
  int abase[4];  // lane's A base (clamped memory-safe for invalid m)
#pragma unroll
  for (int mi = 0; mi < 4; mi++) {
    int ml = wr * 64 + mi * 16 + fr;
    int hl = ml / wo_pad;
    int wo = ml - hl * wo_pad;
    if (hl >= rows_pb) hl = rows_pb - 1;   // slack lanes: clamp (discarded)
    abase[mi] = hl * 2 * xpitch + wo * 6 + fk * 8;
  }
#pragma unroll
  for (int r = 0; r < 7; r++) {
    bf16x8 af[4], bf[2];
#pragma unroll
    for (int mi = 0; mi < 4; mi++) {
      // 4-byte-aligned LDS read path (base is even-element)
      const uint32_t* xu = (const uint32_t*)&Xl[abase[mi] + r * xpitch];
      union { uint32_t u[4]; bf16x8 v; } a;
      a.u[0] = xu[0];
      a.u[1] = xu[1];
      a.u[2] = xu[2];
      a.u[3] = xu[3];
      af[mi] = a.v;
    }
#pragma unroll
    for (int ni = 0; ni < 2; ni++) {
      int ch = wc * 32 + ni * 16 + fr;
      bf[ni] = *(const bf16x8*)&Wl[ch * STEM_WPITCH + r * 32 + fk * 8];
    }
#pragma unroll
    for (int mi = 0; mi < 4; mi++)
#pragma unroll
      for (int ni = 0; ni < 2; ni++)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
  }

  // --- epilogue: y [m][K] + optional BN batch stats ---------------------
  float ssum[2] = {0.f, 0.f}, ssq[2] = {0.f, 0.f};
#pragma unroll
  for (int mi = 0; mi < 4; mi++) {
#pragma unroll
    for (int ni = 0; ni < 2; ni++) {
      int gn = wc * 32 + ni * 16 + fr;
#pragma unroll
      for (int q = 0; q < 4; q++) {
        int ml = wr * 64 + mi * 16 + fk * 4 + q;  // D-frag row = fk*4+q
        int hl = ml / wo_pad;
        int wo = ml - hl * wo_pad;
        if (hl < rows_pb && ho0 + hl < p.Ho && wo < p.Wo && gn < p.K) {
          float v = acc[mi][ni][q];
          long gm = (long)(n * p.Ho + ho0 + hl) * p.Wo + wo;
          Y[gm * p.K + gn] = f2b(v);
          if (STATS) {
            ssum[ni] += v;
            ssq[ni] += v * v;
          }
        }
      }
    }
  }
  if (STATS && stats != nullptr) {
#pragma unroll
    for (int ni = 0; ni < 2; ni++) {
      float s = ssum[ni] + __shfl_xor(ssum[ni], 16, 64);
      s += __shfl_xor(s, 32, 64);
      float s2 = ssq[ni] + __shfl_xor(ssq[ni], 16, 64);
      s2 += __shfl_xor(s2, 32, 64);
      int gn = wc * 32 + ni * 16 + fr;
      if (fk == 0 && gn < p.K) {
        atomicAdd(&stats[gn], s);
        atomicAdd(&stats[p.K + gn], s2);
      }
    }
  }
}

static inline bool stem_eligible(const ConvP& p) {
  // measured SLOWER than the generic t128x64 im2col path everywhere
  // (CIFAR 31.5 vs 30.5 us incl. BN; @224 238 vs 206): per-block W LDS
  // staging (64x232 padded image re-built 3.5k times) eats the gather
  // savings.  Kept as an opt-in experiment + measured negative
  // (docs/STATUS.md); default OFF.
  static int on = [] {
    const char* e = getenv("HZ_STEM_DIRECT");
    return e ? atoi(e) : 0;
  }();
  return on && p.C == 3 && p.R == 7 && p.S == 7 && p.str == 2 &&
         p.pad == 3 && p.K <= 64 && p.Wo <= 128;
}

static void launch_stem_conv(const bf16* x, const bf16* w, bf16* y,
                             float* stats, const ConvP& p, hipStream_t st) {
  int wo_pad = ((p.Wo + 15) / 16) * 16;
  int rows_pb = 128 / wo_pad;
  if (rows_pb < 1) rows_pb = 1;
  int bpi = (p.Ho + rows_pb - 1) / rows_pb;
  int amax = 6 * (wo_pad - 1) + 33;
  int xp = 9 + 3 * p.W + 8;
  if (xp < amax) xp = amax;
  xp = ((xp + 7) / 8) * 8;
  int nrows = rows_pb * 2 + 5;
  size_t lds = (size_t)(64 * STEM_WPITCH + nrows * xp) * sizeof(bf16);
  dim3 grid(p.Nb * bpi);
  if (stats != nullptr)
    k_stem_conv<true><<<grid, 256, lds, st>>>(x, w, y, stats, p, rows_pb,
                                              wo_pad, xp, bpi);
  else
    k_stem_conv<false><<<grid, 256, lds, st>>>(x, w, y, nullptr, p, rows_pb,
                                               wo_pad, xp, bpi);
}

// Throughput-tile selection: the biggest block tile whose grid still
// fills the 256 CUs (HZ_TILE_FILL, default 192 blocks).  CIFAR-shape
// grids stay on the latency-optimized 64×64 tile; ImageNet-shaped convs
// (ResNet50@224 etc.) move to 128×64 / 128×128 where each wave issues
// 2–4× the MFMAs per LDS fragment read.
static inline int pick_tile_env(int M, int N, int fill, int m_min,
                                int m_min256) {
  if (M < m_min) return 0;
  if (M >= m_min256 && N < 128
      && (long)cdiv_h(M, 256) * cdiv_h(N, 64) >= fill)
    return 3;
  if (N >= 128 && (long)cdiv_h(M, 128) * cdiv_h(N, 128) >= fill) return 2;
  if ((long)cdiv_h(M, 128) * cdiv_h(N, 64) >= fill) return 1;
  return 0;
}

// dgrad has its own thresholds: no stats epilogue and an RSCK B-gather,
// so the fwd-measured floors need not match (HZ_TILE_*_DG)
static inline int pick_tile_dgrad(int M, int N) {
  static int fill = [] {
    const char* e = getenv("HZ_TILE_FILL_DG");
    return e ? atoi(e) : 192;
  }();
  static int m_min = [] {
    const char* e = getenv("HZ_TILE_M_MIN_DG");
    return e ? atoi(e) : 16384;
  }();
  static int m256 = [] {
    const char* e = getenv("HZ_TILE_M256_DG");
    return e ? atoi(e) : 65536;
  }();
  return pick_tile_env(M, N, fill, m_min, m256);
}

static inline int pick_tile(int M, int N) {
  static int fill = [] {
    const char* e = getenv("HZ_TILE_FILL");
    return e ? atoi(e) : 192;
  }();
  // measured (gpurun conv sweep, r50@224 bs=32): the 128-row tiles win
  // from M≈25k up (+30-60% fwd) but lose ~20% at M≈6k even when the
  // grid-fill bound is met — keep a hard M floor
  static int m_min = [] {
    const char* e = getenv("HZ_TILE_M_MIN");
    return e ? atoi(e) : 16384;
  }();
  static int m_min256 = [] {
    const char* e = getenv("HZ_TILE_M256");
    return e ? atoi(e) : 65536;
  }();
  if (M < m_min) return 0;
  if (M >= m_min256 && N < 128
      && (long)cdiv_h(M, 256) * cdiv_h(N, 64) >= fill)
    return 3;  // 256x64: very-large-M thin-N shapes (stem/layer1 @224)
  if (N >= 128 && (long)cdiv_h(M, 128) * cdiv_h(N, 128) >= fill) return 2;
  if ((long)cdiv_h(M, 128) * cdiv_h(N, 64) >= fill) return 1;
  return 0;
}

// Non-split forward (bf16 out + fused stats).
extern "C" void launch_conv_fwd(const void* x, const void* w, void* y,
                                float* stats, ConvP p, hipStream_t st) {
  if (stem_eligible(p)) {
    launch_stem_conv((const bf16*)x, (const bf16*)w, (bf16*)y, stats, p,
                     st);
    return;
  }
  bool vec = (p.C % 8) == 0 && (p.Kd % 8) == 0;
  bool s = stats != nullptr;
  auto A = (const bf16*)x;
  auto B = (const bf16*)w;
  auto Y = (bf16*)y;
  MagicP mg = make_magic(p, 1);
  int tile = pick_tile(p.M, p.K);
#define CASE_T(VA, VB, ST, TM, TN)                                     \
  k_conv_mfma<1, VA, VB, ST, false, TM, TN><<<grid, 256, 0, st>>>(     \
      A, B, Y, nullptr, stats, p, mg, p.K, 0, 0)
#define DISPATCH(TM, TN)                                               \
  do {                                                                 \
    dim3 grid(cdiv_h(p.K, TN), cdiv_h(p.M, TM));                       \
    if (vec && s) CASE_T(true, true, true, TM, TN);                    \
    else if (vec) CASE_T(true, true, false, TM, TN);                   \
    else if (s) CASE_T(false, false, true, TM, TN);                    \
    else CASE_T(false, false, false, TM, TN);                          \
  } while (0)
  if (tile == 3) DISPATCH(256, 64);
  else if (tile == 2) DISPATCH(128, 128);
  else if (tile == 1) DISPATCH(128, 64);
  else DISPATCH(64, 64);
#undef DISPATCH
#undef CASE_T
}

// Split-K forward: f32 atomic partials into ws (pre-zeroed [M][K]).
extern "C" void launch_conv_fwd_splitk(const void* x, const void* w,
                                       float* ws, ConvP p, int splitk,
                                       hipStream_t st) {
  int kchunk = cdiv_h(cdiv_h(p.Kd, splitk), 32) * 32;
  splitk = cdiv_h(p.Kd, kchunk);
  bool vec = (p.C % 8) == 0 && (p.Kd % 8) == 0;
  auto A = (const bf16*)x;
  auto B = (const bf16*)w;
  MagicP mg = make_magic(p, 1);
  // fill-split shapes keep their throughput tile; the classic small-M
  // split shapes pick tile 0 (64x64) as before
  int tile = pick_tile(p.M, p.K);
#define SK_T(VA, VB, TM, TN)                                              k_conv_mfma<1, VA, VB, false, true, TM, TN><<<grid, 256, 0, st>>>(          A, B, nullptr, ws, nullptr, p, mg, p.K, kchunk, 0)
#define DISPATCH(TM, TN)                                                  do {                                                                      dim3 grid(cdiv_h(p.K, TN), cdiv_h(p.M, TM), splitk);                    if (vec) SK_T(true, true, TM, TN);                                      else SK_T(false, false, TM, TN);                                      } while (0)
  if (tile == 3) DISPATCH(256, 64);
  else if (tile == 2) DISPATCH(128, 128);
  else if (tile == 1) DISPATCH(128, 64);
  else DISPATCH(64, 64);
#undef DISPATCH
#undef SK_T
}

extern "C" void launch_conv_dgrad(const void* dz, const void* w_rsck,
                                  void* dx, float* ws, int splitk, ConvP p,
                                  int accum, hipStream_t st) {
  // p here: M = Nb*H*W, Kd = R*S*K, output channels = C
  bool vec = (p.K % 8) == 0;
  auto A = (const bf16*)dz;
  auto B = (const bf16*)w_rsck;
  if (splitk > 1) {
    int kchunk = cdiv_h(cdiv_h(p.Kd, splitk), 32) * 32;
    splitk = cdiv_h(p.Kd, kchunk);
    MagicP mg = make_magic(p, 2);
    int tile = pick_tile_dgrad(p.M, p.C);
#define SKD_T(VA, VB, TM, TN)                                             k_conv_mfma<2, VA, VB, false, true, TM, TN><<<grid, 256, 0, st>>>(          A, B, nullptr, ws, nullptr, p, mg, p.C, kchunk, 0)
#define DISPATCH(TM, TN)                                                  do {                                                                      dim3 grid(cdiv_h(p.C, TN), cdiv_h(p.M, TM), splitk);                    if (vec) SKD_T(true, true, TM, TN);                                     else SKD_T(false, false, TM, TN);                                     } while (0)
    if (tile == 3) DISPATCH(256, 64);
    else if (tile == 2) DISPATCH(128, 128);
    else if (tile == 1) DISPATCH(128, 64);
    else DISPATCH(64, 64);
#undef DISPATCH
#undef SKD_T
  } else {
    MagicP mg = make_magic(p, 2);
    int tile = pick_tile_dgrad(p.M, p.C);
#define DGRAD_T(VA, VB, TM, TN)                                         \
  k_conv_mfma<2, VA, VB, false, false, TM, TN><<<grid, 256, 0, st>>>(   \
      A, B, (bf16*)dx, nullptr, nullptr, p, mg, p.C, 0, accum)
#define DISPATCH(TM, TN)                                                \
  do {                                                                  \
    dim3 grid(cdiv_h(p.C, TN), cdiv_h(p.M, TM));                        \
    if (vec) DGRAD_T(true, true, TM, TN);                               \
    else DGRAD_T(false, false, TM, TN);                                 \
  } while (0)
    if (tile == 3) DISPATCH(256, 64);
    else if (tile == 2) DISPATCH(128, 128);
    else if (tile == 1) DISPATCH(128, 64);
    else DISPATCH(64, 64);
#undef DISPATCH
#undef DGRAD_T
  }
}

extern "C" void launch_gemm_bf16(const void* a, const void* b, void* c, int M,
                                 int N, int K, hipStream_t st) {
  ConvP p{};
  p.M = M;
  p.Kd = K;
  dim3 grid(cdiv_h(N, 64), cdiv_h(M, 64));
  bool vec = (K % 8) == 0;
  MagicP mg{};
  if (vec)
    k_conv_mfma<0, true, true, false, false>
        <<<grid, 256, 0, st>>>((const bf16*)a, (const bf16*)b, (bf16*)c,
                               nullptr, nullptr, p, mg, N, 0, 0);
  else
    k_conv_mfma<0, false, false, false, false>
        <<<grid, 256, 0, st>>>((const bf16*)a, (const bf16*)b, (bf16*)c,
                               nullptr, nullptr, p, mg, N, 0, 0);
}

// tile class per task batch: tx/ty counted in (tk3, tko)-wide tiles by
// the caller.  128-wide k3 halves Dz re-reads; 128-wide ko halves X
// re-reads; both double the MFMA per staging write.
extern "C" void launch_wgrad_batched_t(const void* args, int blocks,
                                       int tk3, int tko, hipStream_t st) {
  if (blocks <= 0) return;
  const WgradBatchArgs& a = *(const WgradBatchArgs*)args;
  if (tk3 == 128 && tko == 128)
    k_wgrad_batched<128, 128><<<blocks, 256, 0, st>>>(a);
  else if (tk3 == 128)
    k_wgrad_batched<128, 64><<<blocks, 256, 0, st>>>(a);
  else if (tko == 128)
    k_wgrad_batched<64, 128><<<blocks, 256, 0, st>>>(a);
  else
    k_wgrad_batched<64, 64><<<blocks, 256, 0, st>>>(a);
}

extern "C" void launch_wgrad_reduce_batched(const void* args, int blocks,
                                            hipStream_t st) {
  if (blocks > 0)
    k_wgrad_reduce_batched<<<blocks, 256, 0, st>>>(
        *(const WredBatchArgs*)args);
}

extern "C" int wgrad_msplit(ConvP p) {
  int tiles = cdiv_h(p.Kd, 64) * cdiv_h(p.K, 64);
  int msplit = max(1, min(cdiv_h(p.M, 32), 512 / max(1, tiles)));
  int mchunk = cdiv_h(cdiv_h(p.M, msplit), 32) * 32;
  return cdiv_h(p.M, mchunk);
}

// ws: per-split slab workspace [msplit][K][Kd] f32 (only read/written when
// msplit > 1); dw must be pre-zeroed (direct-grad .grad view or fresh).
extern "C" void launch_wgrad(const void* x, const void* dz, float* dw,
                             float* ws, ConvP p, hipStream_t st) {
  int msplit = wgrad_msplit(p);
  int mchunk = cdiv_h(cdiv_h(p.M, msplit), 32) * 32;
  dim3 grid(cdiv_h(p.Kd, 64), cdiv_h(p.K, 64), msplit);
  bool vec = (p.C % 8) == 0;
  float* out = msplit > 1 ? ws : dw;
  MagicP mg = make_magic(p, 1);
  if (vec)
    k_wgrad<true><<<grid, 256, 0, st>>>((const bf16*)x, (const bf16*)dz, out,
                                        p, mg, p.K, mchunk);
  else
    k_wgrad<false><<<grid, 256, 0, st>>>((const bf16*)x, (const bf16*)dz,
                                         out, p, mg, p.K, mchunk);
  if (msplit > 1) {
    long n = (long)p.K * p.Kd;
    int blocks = (int)min((n / 4 + 255) / 256, (long)1024);
    k_wgrad_reduce<<<blocks, 256, 0, st>>>(ws, dw, n, msplit);
  }
}
