// Implicit-GEMM convolution kernels for gfx950 (MI355X, CDNA4).
//
// One MFMA tile template serves three modes:
//   MODE 0  GEMM      C[M,N]  = A[M,Kd] · B[N,Kd]^T            (bf16 in, bf16 out)
//   MODE 1  CONV_FWD  y[m,ko] = Σ_{r,s,c} x[patch] · W[ko,r,s,c]   (NHWC / KRSC)
//   MODE 2  CONV_DGRAD dx[m,c] = Σ_{r,s,ko} dz[mapped] · Wr[r,s,c,ko] (RSCK)
//
// Geometry: 64×64 block tile, BK=32 K-steps, 256 threads = 4 waves in 2×2,
// each wave computes a 32×32 sub-tile as 2×2 mfma_f32_16x16x32_bf16
// fragments accumulating in f32.  A/B tiles are LDS-staged with +8-element
// row padding (row stride 80 B → 16-lane ds_read_b128 groups land on
// distinct banks).  The forward kernel's epilogue also accumulates the
// per-channel BN batch statistics (Σy, Σy²) with wave-level shuffle
// reduction + one atomicAdd per 16-lane group, so a training conv+BN block
// is 2 kernels total (this + bn_apply) instead of torch's ~5
// (north-star fused conv+BN+ReLU requirement).
#include "common.h"

struct ConvP {
  int Nb, H, W, C, K;      // batch, input spatial, in/out channels
  int Ho, Wo, R, S;        // output spatial, filter
  int str, pad;
  int M, Kd;               // GEMM rows (=Nb*Ho*Wo fwd, Nb*H*W dgrad), reduction size
};

// ---------------------------------------------------------------- staging --
template <int MODE>
DEV int a_addr(const ConvP& p, int m, int k, bool& valid) {
  if (MODE == 0) {
    valid = (m < p.M) & (k < p.Kd);
    return m * p.Kd + k;
  } else if (MODE == 1) {
    // x gather: m -> (n,ho,wo), k -> (r,s,c)
    int n = m / (p.Ho * p.Wo), hw = m % (p.Ho * p.Wo);
    int ho = hw / p.Wo, wo = hw % p.Wo;
    int r = k / (p.S * p.C), rm = k % (p.S * p.C);
    int s = rm / p.C, c = rm % p.C;
    int hi = ho * p.str - p.pad + r, wi = wo * p.str - p.pad + s;
    valid = (m < p.M) & (k < p.Kd) & (hi >= 0) & (hi < p.H) & (wi >= 0) & (wi < p.W);
    return ((n * p.H + hi) * p.W + wi) * p.C + c;
  } else {
    // dz gather (dgrad): m -> (n,hi,wi) over input dims, k -> (r,s,ko)
    int n = m / (p.H * p.W), hw = m % (p.H * p.W);
    int hi = hw / p.W, wi = hw % p.W;
    int r = k / (p.S * p.K), rm = k % (p.S * p.K);
    int s = rm / p.K, ko = rm % p.K;
    int hs = hi + p.pad - r, ws = wi + p.pad - s;
    bool ok = (m < p.M) & (k < p.Kd) & (hs >= 0) & (ws >= 0) &&
              (hs % p.str == 0) && (ws % p.str == 0);
    int ho = hs / p.str, wo = ws / p.str;
    ok = ok && (ho < p.Ho) && (wo < p.Wo);
    valid = ok;
    return ((n * p.Ho + ho) * p.Wo + wo) * p.K + ko;
  }
}

template <int MODE>
DEV int b_addr(const ConvP& p, int n, int k, int Ntot, bool& valid) {
  if (MODE == 2) {
    // W_rsck[(r*S+s)*C + c][ko] with k = (r*S+s)*K + ko, n = c
    int rs = k / p.K, ko = k % p.K;
    valid = (n < Ntot) & (k < p.Kd);
    return (rs * p.C + n) * p.K + ko;
  }
  // MODE 0/1: row-major [Ntot][Kd] (KRSC weights are exactly this for fwd)
  valid = (n < Ntot) & (k < p.Kd);
  return n * p.Kd + k;
}

// Load 8 contiguous-k elements (vector when layout allows, scalar otherwise).
template <int MODE, bool VEC>
DEV void stage8_a(bf16* dst, const bf16* __restrict__ src, const ConvP& p,
                  int m, int k) {
  V8 v;
  if (VEC) {
    bool ok;
    int a = a_addr<MODE>(p, m, k, ok);
    if (ok) v.u = *(const uint4*)(src + a);
    else v.u = uint4{0, 0, 0, 0};
  } else {
    for (int e = 0; e < 8; e++) {
      bool ok;
      int a = a_addr<MODE>(p, m, k + e, ok);
      v.e[e] = ok ? src[a] : (bf16)0.f;
    }
  }
  *(V8*)dst = v;
}

template <int MODE, bool VEC>
DEV void stage8_b(bf16* dst, const bf16* __restrict__ src, const ConvP& p,
                  int n, int k, int Ntot) {
  V8 v;
  if (VEC) {
    bool ok;
    int a = b_addr<MODE>(p, n, k, Ntot, ok);
    if (ok) v.u = *(const uint4*)(src + a);
    else v.u = uint4{0, 0, 0, 0};
  } else {
    for (int e = 0; e < 8; e++) {
      bool ok;
      int a = b_addr<MODE>(p, n, k + e, Ntot, ok);
      v.e[e] = ok ? src[a] : (bf16)0.f;
    }
  }
  *(V8*)dst = v;
}

// ------------------------------------------------------------- main tile --
// LDS row stride 40 elems (32 + 8 pad) = 80 B.
#define LDA 40

template <int MODE, bool VECA, bool VECB, bool STATS>
__global__ __launch_bounds__(256) void k_conv_mfma(
    const bf16* __restrict__ A, const bf16* __restrict__ Bw,
    bf16* __restrict__ Y, float* __restrict__ stats, ConvP p, int Ntot) {
  __shared__ bf16 As[64 * LDA];
  __shared__ bf16 Bs[64 * LDA];

  const int m0 = blockIdx.y * 64, n0 = blockIdx.x * 64;
  const int tid = threadIdx.x;
  const int srow = tid >> 2, scol = (tid & 3) * 8;  // staging: 1 vec8 each
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int fr = lane & 15, fk = lane >> 4;

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < p.Kd; k0 += 32) {
    stage8_a<MODE, VECA>(&As[srow * LDA + scol], A, p, m0 + srow, k0 + scol);
    stage8_b<MODE, VECB>(&Bs[srow * LDA + scol], Bw, p, n0 + srow, k0 + scol,
                         Ntot);
    __syncthreads();
    bf16x8 af[2], bf[2];
#pragma unroll
    for (int mi = 0; mi < 2; mi++)
      af[mi] = *(const bf16x8*)&As[(wr * 32 + mi * 16 + fr) * LDA + fk * 8];
#pragma unroll
    for (int ni = 0; ni < 2; ni++)
      bf[ni] = *(const bf16x8*)&Bs[(wc * 32 + ni * 16 + fr) * LDA + fk * 8];
#pragma unroll
    for (int mi = 0; mi < 2; mi++)
#pragma unroll
      for (int ni = 0; ni < 2; ni++)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
  }

  // Epilogue. D fragment: col = lane&15 (=fr), row = fk*4 + q.
  float ssum[2] = {0.f, 0.f}, ssq[2] = {0.f, 0.f};
#pragma unroll
  for (int mi = 0; mi < 2; mi++) {
#pragma unroll
    for (int ni = 0; ni < 2; ni++) {
      int gn = n0 + wc * 32 + ni * 16 + fr;
#pragma unroll
      for (int q = 0; q < 4; q++) {
        int gm = m0 + wr * 32 + mi * 16 + fk * 4 + q;
        if (gm < p.M && gn < Ntot) {
          float v = acc[mi][ni][q];
          Y[(long)gm * Ntot + gn] = f2b(v);
          if (STATS) {
            ssum[ni] += v;
            ssq[ni] += v * v;
          }
        }
      }
    }
  }
  if (STATS && stats != nullptr) {
    // reduce the 4 lanes sharing a column (l, l+16, l+32, l+48)
#pragma unroll
    for (int ni = 0; ni < 2; ni++) {
      float s = ssum[ni] + __shfl_xor(ssum[ni], 16, 64);
      s += __shfl_xor(s, 32, 64);
      float s2 = ssq[ni] + __shfl_xor(ssq[ni], 16, 64);
      s2 += __shfl_xor(s2, 32, 64);
      int gn = n0 + wc * 32 + ni * 16 + fr;
      if (fk == 0 && gn < Ntot) {
        atomicAdd(&stats[gn], s);
        atomicAdd(&stats[Ntot + gn], s2);
      }
    }
  }
}

// ------------------------------------------------------------------ wgrad --
// dW[ko][k3] += Σ_m patch(m,k3) · dz(m,ko);  dW is KRSC flat [Ntot][Kd] f32
// (identical layout to the f32 master parameter → autograd-direct).
// VALU outer-product v1: 64×64 tile per block, 4×4 f32 accumulators per
// thread, M reduced in 32-deep LDS-staged chunks, msplit-way M parallelism
// with f32 atomics.  (MFMA tr-read upgrade is a follow-up; wgrad ≈ 1/3 of
// backward FLOPs at these shapes.)
#define LDW 72

template <bool VECA>
__global__ __launch_bounds__(256) void k_wgrad(
    const bf16* __restrict__ X, const bf16* __restrict__ Dz,
    float* __restrict__ dW, ConvP p, int Ntot, int mchunk) {
  __shared__ bf16 As[32 * LDW];   // [m within chunk][k3]
  __shared__ bf16 Ds[32 * LDW];   // [m within chunk][ko]
  const int k3_0 = blockIdx.x * 64, n0 = blockIdx.y * 64;
  const int mbeg = blockIdx.z * mchunk;
  const int mend = min(p.M, mbeg + mchunk);
  const int tid = threadIdx.x;
  const int srow = tid >> 3, scol = (tid & 7) * 8;  // 32×64 tile: 1 vec8 each
  const int trow = (tid >> 4) * 4, tcol = (tid & 15) * 4;

  float acc[4][4] = {};
  for (int m0 = mbeg; m0 < mend; m0 += 32) {
    stage8_a<1, VECA>(&As[srow * LDW + scol], X, p, m0 + srow, k3_0 + scol);
    {
      V8 v;
      int m = m0 + srow, n = n0 + scol;
      if (m < p.M && n < Ntot)
        v.u = *(const uint4*)(Dz + (long)m * Ntot + n);
      else
        v.u = uint4{0, 0, 0, 0};
      *(V8*)&Ds[srow * LDW + scol] = v;
    }
    __syncthreads();
#pragma unroll 4
    for (int mm = 0; mm < 32; mm++) {
      float a[4], d[4];
#pragma unroll
      for (int i = 0; i < 4; i++) a[i] = b2f(As[mm * LDW + trow + i]);
#pragma unroll
      for (int j = 0; j < 4; j++) d[j] = b2f(Ds[mm * LDW + tcol + j]);
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++) acc[i][j] = fmaf(a[i], d[j], acc[i][j]);
    }
    __syncthreads();
  }
#pragma unroll
  for (int i = 0; i < 4; i++)
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int k3 = k3_0 + trow + i, n = n0 + tcol + j;
      if (k3 < p.Kd && n < Ntot)
        atomicAdd(&dW[(long)n * p.Kd + k3], acc[i][j]);
    }
}

// ------------------------------------------------------------- launchers --
static inline int cdiv_h(int a, int b) { return (a + b - 1) / b; }

extern "C" void launch_conv_fwd(const void* x, const void* w, void* y,
                                float* stats, ConvP p, hipStream_t st) {
  dim3 grid(cdiv_h(p.K, 64), cdiv_h(p.M, 64));
  bool veca = (p.C % 8) == 0;
  bool vecb = (p.Kd % 8) == 0;
  bool s = stats != nullptr;
  auto A = (const bf16*)x;
  auto B = (const bf16*)w;
  auto Y = (bf16*)y;
#define CASE(VA, VB, ST)                                                   \
  k_conv_mfma<1, VA, VB, ST><<<grid, 256, 0, st>>>(A, B, Y, stats, p, p.K)
  if (veca && vecb && s) CASE(true, true, true);
  else if (veca && vecb) CASE(true, true, false);
  else if (s) CASE(false, false, true);
  else CASE(false, false, false);
#undef CASE
}

extern "C" void launch_conv_dgrad(const void* dz, const void* w_rsck, void* dx,
                                  ConvP p, hipStream_t st) {
  // p here: M = Nb*H*W, Kd = R*S*K, output channels = C
  dim3 grid(cdiv_h(p.C, 64), cdiv_h(p.M, 64));
  bool vec = (p.K % 8) == 0;
  auto A = (const bf16*)dz;
  auto B = (const bf16*)w_rsck;
  auto Y = (bf16*)dx;
  if (vec)
    k_conv_mfma<2, true, true, false>
        <<<grid, 256, 0, st>>>(A, B, Y, nullptr, p, p.C);
  else
    k_conv_mfma<2, false, false, false>
        <<<grid, 256, 0, st>>>(A, B, Y, nullptr, p, p.C);
}

extern "C" void launch_gemm_bf16(const void* a, const void* b, void* c, int M,
                                 int N, int K, hipStream_t st) {
  ConvP p{};
  p.M = M;
  p.Kd = K;
  dim3 grid(cdiv_h(N, 64), cdiv_h(M, 64));
  bool vec = (K % 8) == 0;
  if (vec)
    k_conv_mfma<0, true, true, false>
        <<<grid, 256, 0, st>>>((const bf16*)a, (const bf16*)b, (bf16*)c,
                               nullptr, p, N);
  else
    k_conv_mfma<0, false, false, false>
        <<<grid, 256, 0, st>>>((const bf16*)a, (const bf16*)b, (bf16*)c,
                               nullptr, p, N);
}

extern "C" void launch_wgrad(const void* x, const void* dz, float* dw, ConvP p,
                             hipStream_t st) {
  int tiles = cdiv_h(p.Kd, 64) * cdiv_h(p.K, 64);
  int msplit = max(1, min(cdiv_h(p.M, 32), 256 / max(1, tiles)));
  int mchunk = cdiv_h(cdiv_h(p.M, msplit), 32) * 32;
  msplit = cdiv_h(p.M, mchunk);
  dim3 grid(cdiv_h(p.Kd, 64), cdiv_h(p.K, 64), msplit);
  bool vec = (p.C % 8) == 0;
  if (vec)
    k_wgrad<true><<<grid, 256, 0, st>>>((const bf16*)x, (const bf16*)dz, dw, p,
                                        p.K, mchunk);
  else
    k_wgrad<false><<<grid, 256, 0, st>>>((const bf16*)x, (const bf16*)dz, dw,
                                         p, p.K, mchunk);
}
