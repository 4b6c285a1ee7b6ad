// Common device helpers for the horizonml_amd gfx950 kernels.
// CDNA4-only: wave64, MFMA bf16, LDS-staged tiles. No CUDA compat paths.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;

using bf16 = __bf16;

DEV float b2f(bf16 x) { return (float)x; }
DEV bf16 f2b(float x) { return (bf16)x; }   // clang emits v_cvt (RNE) on gfx950

// 16-byte vector of 8 bf16 for global loads/stores.
union V8 {
  bf16x8 v;
  uint4 u;
  bf16 e[8];
};

DEV int cdiv(int a, int b) { return (a + b - 1) / b; }

// Block-wide f32 reduction helper over one wave via xor shuffles (wave64).
DEV float wave_reduce_sum(float x) {
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_xor(x, off, 64);
  return x;
}

// Fast division by a runtime constant (cutlass FastDivmod form):
// q = (umulhi(n, m) + n) >> s for 1 < d < 2^31, 0 <= n < 2^31; d == 1 is
// the identity.  The implicit-GEMM gathers decompose linear indices with
// 4-5 divisions per vec8 — emulated u32 division costs ~25-40 VALU each.
struct MagicP {
  unsigned m[4];
  int s[4];
  unsigned d[4];
};

DEV unsigned fdiv(unsigned n, const MagicP& mg, int i) {
  return mg.d[i] == 1 ? n : (__umulhi(n, mg.m[i]) + n) >> mg.s[i];
}

DEV unsigned fmod(unsigned n, unsigned q, const MagicP& mg, int i) {
  return n - q * mg.d[i];
}
