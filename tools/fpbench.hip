// isolated fp_mul variant microbench (not part of the product build)
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include "bls_consts.h"

struct fp { uint64_t v[6]; };

__device__ __forceinline__ bool fp_ge_p(const uint64_t t[6]) {
#pragma unroll
  for (int i = 5; i >= 0; i--) { if (t[i] > BLS_P[i]) return true; if (t[i] < BLS_P[i]) return false; }
  return true;
}
__device__ __forceinline__ void fp_sub_p(uint64_t t[6]) {
  unsigned __int128 bw = 0;
#pragma unroll
  for (int i = 0; i < 6; i++) { unsigned __int128 x = (unsigned __int128)t[i] - BLS_P[i] - (uint64_t)bw; t[i] = (uint64_t)x; bw = (x >> 64) & 1; }
}
// variant A: CIOS
__device__ __forceinline__ void mulA(fp &r, const fp &a, const fp &b) {
  uint64_t t[8];
#pragma unroll
  for (int i = 0; i < 8; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    unsigned __int128 c = 0;
#pragma unroll
    for (int j = 0; j < 6; j++) { c += (unsigned __int128)a.v[j] * b.v[i] + t[j]; t[j] = (uint64_t)c; c >>= 64; }
    c += t[6]; t[6] = (uint64_t)c; t[7] = (uint64_t)(c >> 64);
    uint64_t m = t[0] * BLS_N0;
    c = ((unsigned __int128)m * BLS_P[0] + t[0]) >> 64;
#pragma unroll
    for (int j = 1; j < 6; j++) { c += (unsigned __int128)m * BLS_P[j] + t[j]; t[j-1] = (uint64_t)c; c >>= 64; }
    c += t[6]; t[5] = (uint64_t)c; t[6] = t[7] + (uint64_t)(c >> 64); t[7] = 0;
  }
  if (t[6] || fp_ge_p(t)) fp_sub_p(t);
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = t[i];
}
// variant B: column scan + SOS reduce
__device__ __forceinline__ void mulB(fp &r, const fp &a, const fp &b) {
  uint64_t t[13];
  uint64_t lo = 0, mid = 0, hi = 0;
#pragma unroll
  for (int k = 0; k < 11; k++) {
    const int i0 = k < 6 ? 0 : k - 5;
    const int i1 = k < 6 ? k : 5;
#pragma unroll
    for (int i = i0; i <= i1; i++) {
      uint64_t pl = a.v[i] * b.v[k - i];
      uint64_t ph = __umul64hi(a.v[i], b.v[k - i]);
      lo += pl; uint64_t c1 = lo < pl;
      mid += c1; hi += (mid < c1);
      mid += ph; hi += (mid < ph);
    }
    t[k] = lo; lo = mid; mid = hi; hi = 0;
  }
  t[11] = lo; t[12] = 0;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    uint64_t m = t[i] * BLS_N0;
    unsigned __int128 c = ((unsigned __int128)m * BLS_P[0] + t[i]) >> 64;
#pragma unroll
    for (int j = 1; j < 6; j++) { c += (unsigned __int128)m * BLS_P[j] + t[i+j]; t[i+j] = (uint64_t)c; c >>= 64; }
#pragma unroll
    for (int j = i + 6; j < 13; j++) { c += t[j]; t[j] = (uint64_t)c; c >>= 64; }
  }
  uint64_t res[6];
#pragma unroll
  for (int i = 0; i < 6; i++) res[i] = t[6 + i];
  if (t[12] || fp_ge_p(res)) fp_sub_p(res);
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = res[i];
}

// variant C: dependent chain of Karatsuba fp2 multiplies (3 INDEPENDENT
// fp_muls per step) — measures whether the compiler interleaves them
struct fp2 { fp c0, c1; };
__device__ __forceinline__ void fp_add(fp &r, const fp &a, const fp &b) {
  unsigned __int128 c = 0;
  uint64_t t[6];
#pragma unroll
  for (int i = 0; i < 6; i++) { c += (unsigned __int128)a.v[i] + b.v[i]; t[i] = (uint64_t)c; c >>= 64; }
  if (c || fp_ge_p(t)) fp_sub_p(t);
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = t[i];
}
__device__ __forceinline__ void fp_sub(fp &r, const fp &a, const fp &b) {
  unsigned __int128 bw = 0;
  uint64_t t[6];
#pragma unroll
  for (int i = 0; i < 6; i++) { unsigned __int128 x = (unsigned __int128)a.v[i] - b.v[i] - (uint64_t)bw; t[i] = (uint64_t)x; bw = (x >> 64) & 1; }
  if (bw) { unsigned __int128 c = 0;
#pragma unroll
    for (int i = 0; i < 6; i++) { c += (unsigned __int128)t[i] + BLS_P[i]; t[i] = (uint64_t)c; c >>= 64; } }
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = t[i];
}
__device__ __forceinline__ void mul2(fp2 &r, const fp2 &a, const fp2 &b) {
  fp t0, t1, s0, s1, m;
  mulA(t0, a.c0, b.c0);
  mulA(t1, a.c1, b.c1);
  fp_add(s0, a.c0, a.c1);
  fp_add(s1, b.c0, b.c1);
  mulA(m, s0, s1);
  fp_sub(m, m, t0);
  fp_sub(m, m, t1);
  fp_sub(r.c0, t0, t1);
  r.c1 = m;
}
__global__ void chain2(const uint64_t *in, uint64_t *out, int iters) {
  fp2 a, b;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    a.c0.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; a.c1.v[i] = in[6+i];
    b.c0.v[i] = in[6+i] ^ 0x5555; b.c1.v[i] = in[i] ^ 0x3333;
  }
  b.c0.v[5] %= BLS_P[5]; b.c1.v[5] %= BLS_P[5]; a.c1.v[5] %= BLS_P[5];
  for (int q = 0; q < iters; q++) mul2(a, a, b);
  uint64_t *o = out + 12 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) { o[i] = a.c0.v[i]; o[6+i] = a.c1.v[i]; }
}

// variant D: source-interleaved DUAL CIOS — two independent multiplies in
// one function, statements alternating, so the local scheduler overlaps
// the two dependency chains.
__device__ __forceinline__ void mulA2(fp &r1, const fp &a1, const fp &b1,
                                      fp &r2, const fp &a2, const fp &b2) {
  uint64_t t[8], u[8];
#pragma unroll
  for (int i = 0; i < 8; i++) { t[i] = 0; u[i] = 0; }
#pragma unroll
  for (int i = 0; i < 6; i++) {
    unsigned __int128 c = 0, d = 0;
#pragma unroll
    for (int j = 0; j < 6; j++) {
      c += (unsigned __int128)a1.v[j] * b1.v[i] + t[j];
      d += (unsigned __int128)a2.v[j] * b2.v[i] + u[j];
      t[j] = (uint64_t)c; c >>= 64;
      u[j] = (uint64_t)d; d >>= 64;
    }
    c += t[6]; t[6] = (uint64_t)c; t[7] = (uint64_t)(c >> 64);
    d += u[6]; u[6] = (uint64_t)d; u[7] = (uint64_t)(d >> 64);
    uint64_t m1 = t[0] * BLS_N0;
    uint64_t m2 = u[0] * BLS_N0;
    c = ((unsigned __int128)m1 * BLS_P[0] + t[0]) >> 64;
    d = ((unsigned __int128)m2 * BLS_P[0] + u[0]) >> 64;
#pragma unroll
    for (int j = 1; j < 6; j++) {
      c += (unsigned __int128)m1 * BLS_P[j] + t[j];
      d += (unsigned __int128)m2 * BLS_P[j] + u[j];
      t[j - 1] = (uint64_t)c; c >>= 64;
      u[j - 1] = (uint64_t)d; d >>= 64;
    }
    c += t[6]; t[5] = (uint64_t)c; t[6] = t[7] + (uint64_t)(c >> 64); t[7] = 0;
    d += u[6]; u[5] = (uint64_t)d; u[6] = u[7] + (uint64_t)(d >> 64); u[7] = 0;
  }
  if (t[6] || fp_ge_p(t)) fp_sub_p(t);
  if (u[6] || fp_ge_p(u)) fp_sub_p(u);
#pragma unroll
  for (int i = 0; i < 6; i++) { r1.v[i] = t[i]; r2.v[i] = u[i]; }
}
__global__ void chainD(const uint64_t *in, uint64_t *out, int iters) {
  fp a1, b1, a2, b2;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    a1.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; b1.v[i] = in[6 + i];
    a2.v[i] = in[6 + i] ^ 0x77; b2.v[i] = in[i] ^ 0x99;
  }
  a2.v[5] %= BLS_P[5]; b2.v[5] %= BLS_P[5];
  for (int q = 0; q < iters; q++) mulA2(a1, a1, b1, a2, a2, b2);
  uint64_t *o = out + 12 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) { o[i] = a1.v[i]; o[6 + i] = a2.v[i]; }
}

// variant E: 12x32-bit limb CIOS — each inner mac is one v_mad_u64_u32
// (32x32+64) plus one 64-bit add; the fp layout (LE u64 limbs) aliases
// to LE u32 limbs for free.
__device__ __forceinline__ void mulE(fp &r, const fp &a, const fp &b) {
  const uint32_t *A = reinterpret_cast<const uint32_t *>(a.v);
  const uint32_t *B = reinterpret_cast<const uint32_t *>(b.v);
  uint32_t P32[12];
#pragma unroll
  for (int i = 0; i < 6; i++) {
    P32[2 * i] = (uint32_t)BLS_P[i];
    P32[2 * i + 1] = (uint32_t)(BLS_P[i] >> 32);
  }
  const uint32_t N0_32 = (uint32_t)BLS_N0; // -p^-1 mod 2^32 = low word
  uint32_t t[14];
#pragma unroll
  for (int i = 0; i < 14; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < 12; i++) {
    uint64_t c = 0;
#pragma unroll
    for (int j = 0; j < 12; j++) {
      uint64_t s = (uint64_t)A[j] * B[i] + t[j] + (uint32_t)c;
      // carry (c>>32) folded next step: keep full add for correctness
      s += (c >> 32) ? 0ull : 0ull; // (c fits 32 bits by construction below)
      t[j] = (uint32_t)s;
      c = s >> 32;
    }
    uint64_t s = (uint64_t)t[12] + c;
    t[12] = (uint32_t)s;
    t[13] = (uint32_t)(s >> 32);
    uint32_t m = t[0] * N0_32;
    c = ((uint64_t)m * P32[0] + t[0]) >> 32;
#pragma unroll
    for (int j = 1; j < 12; j++) {
      uint64_t s2 = (uint64_t)m * P32[j] + t[j] + (uint32_t)c;
      t[j - 1] = (uint32_t)s2;
      c = s2 >> 32;
    }
    s = (uint64_t)t[12] + c;
    t[11] = (uint32_t)s;
    t[12] = t[13] + (uint32_t)(s >> 32);
    t[13] = 0;
  }
  uint64_t res[6];
#pragma unroll
  for (int i = 0; i < 6; i++) res[i] = (uint64_t)t[2 * i] | ((uint64_t)t[2 * i + 1] << 32);
  if (t[12] || fp_ge_p(res)) fp_sub_p(res);
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = res[i];
}
__global__ void chainE(const uint64_t *in, uint64_t *out, int iters) {
  fp a, b;
#pragma unroll
  for (int i = 0; i < 6; i++) { a.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; b.v[i] = in[6 + i]; }
  for (int q = 0; q < iters; q++) mulE(a, a, b);
  uint64_t *o = out + 6 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) o[i] = a.v[i];
}

// variant F: dual source-interleaved 12x32 CIOS
__device__ __forceinline__ void mulE2(fp &r1, const fp &a1, const fp &b1,
                                      fp &r2, const fp &a2, const fp &b2) {
  const uint32_t *A1 = reinterpret_cast<const uint32_t *>(a1.v);
  const uint32_t *B1 = reinterpret_cast<const uint32_t *>(b1.v);
  const uint32_t *A2 = reinterpret_cast<const uint32_t *>(a2.v);
  const uint32_t *B2 = reinterpret_cast<const uint32_t *>(b2.v);
  uint32_t P32[12];
#pragma unroll
  for (int i = 0; i < 6; i++) {
    P32[2 * i] = (uint32_t)BLS_P[i];
    P32[2 * i + 1] = (uint32_t)(BLS_P[i] >> 32);
  }
  const uint32_t N0_32 = (uint32_t)BLS_N0;
  uint32_t t[14], u[14];
#pragma unroll
  for (int i = 0; i < 14; i++) { t[i] = 0; u[i] = 0; }
#pragma unroll
  for (int i = 0; i < 12; i++) {
    uint64_t c = 0, d = 0;
#pragma unroll
    for (int j = 0; j < 12; j++) {
      uint64_t s1 = (uint64_t)A1[j] * B1[i] + t[j] + (uint32_t)c;
      uint64_t s2 = (uint64_t)A2[j] * B2[i] + u[j] + (uint32_t)d;
      t[j] = (uint32_t)s1; c = s1 >> 32;
      u[j] = (uint32_t)s2; d = s2 >> 32;
    }
    uint64_t s1 = (uint64_t)t[12] + c; t[12] = (uint32_t)s1; t[13] = (uint32_t)(s1 >> 32);
    uint64_t s2 = (uint64_t)u[12] + d; u[12] = (uint32_t)s2; u[13] = (uint32_t)(s2 >> 32);
    uint32_t m1 = t[0] * N0_32;
    uint32_t m2 = u[0] * N0_32;
    c = ((uint64_t)m1 * P32[0] + t[0]) >> 32;
    d = ((uint64_t)m2 * P32[0] + u[0]) >> 32;
#pragma unroll
    for (int j = 1; j < 12; j++) {
      uint64_t q1 = (uint64_t)m1 * P32[j] + t[j] + (uint32_t)c;
      uint64_t q2 = (uint64_t)m2 * P32[j] + u[j] + (uint32_t)d;
      t[j - 1] = (uint32_t)q1; c = q1 >> 32;
      u[j - 1] = (uint32_t)q2; d = q2 >> 32;
    }
    s1 = (uint64_t)t[12] + c; t[11] = (uint32_t)s1; t[12] = t[13] + (uint32_t)(s1 >> 32); t[13] = 0;
    s2 = (uint64_t)u[12] + d; u[11] = (uint32_t)s2; u[12] = u[13] + (uint32_t)(s2 >> 32); u[13] = 0;
  }
  uint64_t res1[6], res2[6];
#pragma unroll
  for (int i = 0; i < 6; i++) {
    res1[i] = (uint64_t)t[2 * i] | ((uint64_t)t[2 * i + 1] << 32);
    res2[i] = (uint64_t)u[2 * i] | ((uint64_t)u[2 * i + 1] << 32);
  }
  if (t[12] || fp_ge_p(res1)) fp_sub_p(res1);
  if (u[12] || fp_ge_p(res2)) fp_sub_p(res2);
#pragma unroll
  for (int i = 0; i < 6; i++) { r1.v[i] = res1[i]; r2.v[i] = res2[i]; }
}
__global__ void chainF(const uint64_t *in, uint64_t *out, int iters) {
  fp a1, b1, a2, b2;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    a1.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; b1.v[i] = in[6 + i];
    a2.v[i] = in[6 + i] ^ 0x77; b2.v[i] = in[i] ^ 0x99;
  }
  a2.v[5] %= BLS_P[5]; b2.v[5] %= BLS_P[5];
  for (int q = 0; q < iters; q++) mulE2(a1, a1, b1, a2, a2, b2);
  uint64_t *o = out + 12 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) { o[i] = a1.v[i]; o[6 + i] = a2.v[i]; }
}

// ======================= round-2 variants =======================
// PP2 = 2*p^2 as 24 u32 limbs (lazy-reduction offset), compile-time
struct pp2_t { uint32_t w[24]; };
constexpr pp2_t make_pp2() {
  pp2_t r{};
  uint64_t t[13] = {};
  for (int i = 0; i < 6; i++) {
    uint64_t carry = 0;
    for (int j = 0; j < 6; j++) {
      unsigned __int128 acc =
          (unsigned __int128)BLS_P[i] * BLS_P[j] + t[i + j] + carry;
      t[i + j] = (uint64_t)acc;
      carry = (uint64_t)(acc >> 64);
    }
    // propagate into the (fresh-enough) tail
    int k = i + 6;
    while (carry) {
      unsigned __int128 acc = (unsigned __int128)t[k] + carry;
      t[k] = (uint64_t)acc;
      carry = (uint64_t)(acc >> 64);
      k++;
    }
  }
  uint64_t c = 0;
  for (int i = 0; i < 12; i++) {
    uint64_t v = (t[i] << 1) | c;
    c = t[i] >> 63;
    r.w[2 * i] = (uint32_t)v;
    r.w[2 * i + 1] = (uint32_t)(v >> 32);
  }
  return r;
}
__constant__ constexpr pp2_t PP2 = make_pp2();

__device__ __forceinline__ void mul_wide32(uint32_t t[25], const fp &a,
                                           const fp &b) {
  const uint32_t *A = reinterpret_cast<const uint32_t *>(a.v);
  const uint32_t *B = reinterpret_cast<const uint32_t *>(b.v);
#pragma unroll
  for (int i = 0; i < 25; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < 12; i++) {
    uint64_t c = 0;
#pragma unroll
    for (int j = 0; j < 12; j++) {
      uint64_t s = (uint64_t)A[j] * B[i] + t[i + j] + (uint32_t)c;
      t[i + j] = (uint32_t)s;
      c = s >> 32;
    }
    t[i + 12] = (uint32_t)c;
  }
}

// Montgomery reduce a 24-limb T < 3p^2 (< p*2^384): r = T*R^-1 mod p
__device__ __forceinline__ void redc32(fp &r, const uint32_t *T) {
  uint32_t P32[12];
#pragma unroll
  for (int i = 0; i < 6; i++) {
    P32[2 * i] = (uint32_t)BLS_P[i];
    P32[2 * i + 1] = (uint32_t)(BLS_P[i] >> 32);
  }
  const uint32_t N0_32 = (uint32_t)BLS_N0;
  uint32_t t[12];
#pragma unroll
  for (int i = 0; i < 12; i++) t[i] = T[i];
  uint32_t carry = 0;
#pragma unroll
  for (int i = 0; i < 12; i++) {
    uint32_t m = t[0] * N0_32;
    uint64_t c = ((uint64_t)m * P32[0] + t[0]) >> 32;
#pragma unroll
    for (int j = 1; j < 12; j++) {
      uint64_t s = (uint64_t)m * P32[j] + t[j] + (uint32_t)c;
      t[j - 1] = (uint32_t)s;
      c = s >> 32;
    }
    uint64_t s = (uint64_t)T[12 + i] + c + carry;
    t[11] = (uint32_t)s;
    carry = (uint32_t)(s >> 32);
  }
  uint64_t res[6];
#pragma unroll
  for (int i = 0; i < 6; i++)
    res[i] = (uint64_t)t[2 * i] | ((uint64_t)t[2 * i + 1] << 32);
  if (carry || fp_ge_p(res)) fp_sub_p(res);
#pragma unroll
  for (int i = 0; i < 6; i++) r.v[i] = res[i];
}

// r = a + PP2 - b over 24 limbs (keeps the difference positive)
__device__ __forceinline__ void wide_sub_pp2(uint32_t r[25],
                                             const uint32_t a[25],
                                             const uint32_t b[25]) {
  uint64_t c = 0;
  uint32_t tmp[24];
#pragma unroll
  for (int i = 0; i < 24; i++) {
    c += (uint64_t)a[i] + PP2.w[i];
    tmp[i] = (uint32_t)c;
    c >>= 32;
  }
  int64_t bw = 0;
#pragma unroll
  for (int i = 0; i < 24; i++) {
    int64_t x = (int64_t)(uint64_t)tmp[i] - b[i] - bw;
    r[i] = (uint32_t)x;
    bw = (x < 0);
  }
}

__device__ __forceinline__ void wide_add(uint32_t r[25], const uint32_t a[25],
                                         const uint32_t b[25]) {
  uint64_t c = 0;
#pragma unroll
  for (int i = 0; i < 24; i++) {
    c += (uint64_t)a[i] + b[i];
    r[i] = (uint32_t)c;
    c >>= 32;
  }
}

// lazy-reduction Karatsuba fp2 mul: 3 wide products + 2 reductions
__device__ __forceinline__ void mul2_lazy(fp2 &r, const fp2 &a,
                                          const fp2 &b) {
  uint32_t t0[25], t1[25], m[25], s01[25], cw[25];
  fp sa, sb;
  mul_wide32(t0, a.c0, b.c0);
  mul_wide32(t1, a.c1, b.c1);
  fp_add(sa, a.c0, a.c1);
  fp_add(sb, b.c0, b.c1);
  mul_wide32(m, sa, sb);
  wide_sub_pp2(cw, t0, t1);
  redc32(r.c0, cw);
  wide_add(s01, t0, t1);
  wide_sub_pp2(cw, m, s01);
  redc32(r.c1, cw);
}

__global__ void chainG(const uint64_t *in, uint64_t *out, int iters) {
  fp2 a, b;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    a.c0.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; a.c1.v[i] = in[6+i];
    b.c0.v[i] = in[6+i] ^ 0x5555; b.c1.v[i] = in[i] ^ 0x3333;
  }
  b.c0.v[5] %= BLS_P[5]; b.c1.v[5] %= BLS_P[5]; a.c1.v[5] %= BLS_P[5];
  for (int q = 0; q < iters; q++) mul2_lazy(a, a, b);
  uint64_t *o = out + 12 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) { o[i] = a.c0.v[i]; o[6+i] = a.c1.v[i]; }
}

// baseline for G: same fp2 chain with the production 12x32 CIOS
__device__ __forceinline__ void mul2E(fp2 &r, const fp2 &a, const fp2 &b) {
  fp t0, t1, s0, s1, m;
  mulE(t0, a.c0, b.c0);
  mulE(t1, a.c1, b.c1);
  fp_add(s0, a.c0, a.c1);
  fp_add(s1, b.c0, b.c1);
  mulE(m, s0, s1);
  fp_sub(m, m, t0);
  fp_sub(m, m, t1);
  fp_sub(r.c0, t0, t1);
  r.c1 = m;
}
__global__ void chainG0(const uint64_t *in, uint64_t *out, int iters) {
  fp2 a, b;
#pragma unroll
  for (int i = 0; i < 6; i++) {
    a.c0.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; a.c1.v[i] = in[6+i];
    b.c0.v[i] = in[6+i] ^ 0x5555; b.c1.v[i] = in[i] ^ 0x3333;
  }
  b.c0.v[5] %= BLS_P[5]; b.c1.v[5] %= BLS_P[5]; a.c1.v[5] %= BLS_P[5];
  for (int q = 0; q < iters; q++) mul2E(a, a, b);
  uint64_t *o = out + 12 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) { o[i] = a.c0.v[i]; o[6+i] = a.c1.v[i]; }
}

// specialized 12x32 SOS squaring: 66 cross + 12 diag MACs (+ redc 144)
// vs 144+144 for mulE(a,a)
__device__ __forceinline__ void sqr_wide32(uint32_t t[25], const fp &a) {
  const uint32_t *A = reinterpret_cast<const uint32_t *>(a.v);
#pragma unroll
  for (int i = 0; i < 25; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < 11; i++) {
    uint64_t c = 0;
#pragma unroll
    for (int j = i + 1; j < 12; j++) {
      uint64_t s = (uint64_t)A[i] * A[j] + t[i + j] + (uint32_t)c;
      t[i + j] = (uint32_t)s;
      c = s >> 32;
    }
    t[i + 12] = (uint32_t)c;
  }
  uint32_t cc = 0;
#pragma unroll
  for (int i = 0; i < 24; i++) {
    uint32_t nv = (t[i] << 1) | cc;
    cc = t[i] >> 31;
    t[i] = nv;
  }
  uint64_t c = 0;
#pragma unroll
  for (int i = 0; i < 12; i++) {
    uint64_t s = (uint64_t)A[i] * A[i] + t[2 * i] + (uint32_t)c;
    t[2 * i] = (uint32_t)s;
    uint64_t s2 = (uint64_t)t[2 * i + 1] + (s >> 32);
    t[2 * i + 1] = (uint32_t)s2;
    c = s2 >> 32;
  }
}
__device__ __forceinline__ void sqrH(fp &r, const fp &a) {
  uint32_t t[25];
  sqr_wide32(t, a);
  redc32(r, t);
}
template <int USE_SQR>
__global__ void chainH(const uint64_t *in, uint64_t *out, int iters) {
  fp a;
#pragma unroll
  for (int i = 0; i < 6; i++) a.v[i] = in[i] + threadIdx.x + blockIdx.x * 64;
  for (int q = 0; q < iters; q++) {
    if (USE_SQR) sqrH(a, a);
    else mulE(a, a, a);
  }
  uint64_t *o = out + 6 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) o[i] = a.v[i];
}

template <int V>
__global__ void chain(const uint64_t *in, uint64_t *out, int iters) {
  fp a, b;
#pragma unroll
  for (int i = 0; i < 6; i++) { a.v[i] = in[i] + threadIdx.x + blockIdx.x * 64; b.v[i] = in[6 + i]; }
  for (int q = 0; q < iters; q++) {
    if (V == 0) mulA(a, a, b); else mulB(a, a, b);
  }
  uint64_t *o = out + 6 * (blockIdx.x * 64 + threadIdx.x);
#pragma unroll
  for (int i = 0; i < 6; i++) o[i] = a.v[i];
}

int main() {
  uint64_t h_in[12];
  for (int i = 0; i < 12; i++) h_in[i] = 0x123456789abcdefULL * (i + 1);
  h_in[5] %= BLS_P[5]; h_in[11] %= BLS_P[5];
  uint64_t *d_in, *d_out;
  hipMalloc(&d_in, sizeof(h_in));
  hipMalloc(&d_out, 1024 * 64 * 96);
  hipMemcpy(d_in, h_in, sizeof(h_in), hipMemcpyHostToDevice);
  const int ITERS = 20000;
  for (int v = 0; v < 2; v++) {
    // correctness cross-check at 64 iters first
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      if (v == 0) hipLaunchKernelGGL(chain<0>, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      else        hipLaunchKernelGGL(chain<1>, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("variant %c rep %d: %.3f ms (%s) -> %.0f cyc/mul @2.4GHz\n",
             'A' + v, rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS);
    }
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      hipLaunchKernelGGL(chain2, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("variant C (fp2, 3 indep muls) rep %d: %.3f ms (%s) -> %.0f cyc/fp2mul, %.0f cyc per fp_mul-equiv\n",
             rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS, ms * 1e-3 * 2.4e9 / ITERS / 3);
    }
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      hipLaunchKernelGGL(chainD, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("variant D (dual interleaved CIOS) rep %d: %.3f ms (%s) -> %.0f cyc per PAIR, %.0f per mul\n",
             rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS, ms * 1e-3 * 2.4e9 / ITERS / 2);
    }
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      hipLaunchKernelGGL(chainE, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("variant E (12x32 limb CIOS) rep %d: %.3f ms (%s) -> %.0f cyc/mul\n",
             rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS);
    }
    uint64_t ra[6], re[6];
    hipLaunchKernelGGL(chain<0>, dim3(1), dim3(64), 0, 0, d_in, d_out, 777);
    hipMemcpy(ra, d_out, 48, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(chainE, dim3(1), dim3(64), 0, 0, d_in, d_out, 777);
    hipMemcpy(re, d_out, 48, hipMemcpyDeviceToHost);
    bool ok = true;
    for (int i = 0; i < 6; i++) ok &= (ra[i] == re[i]);
    printf("E == A after 777 chained muls: %s\n", ok ? "YES" : "NO");
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      hipLaunchKernelGGL(chainF, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("variant F (dual 12x32 CIOS) rep %d: %.3f ms (%s) -> %.0f cyc/PAIR, %.0f per mul\n",
             rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS, ms * 1e-3 * 2.4e9 / ITERS / 2);
    }
    uint64_t ra[6], rf[12];
    hipLaunchKernelGGL(chain<0>, dim3(1), dim3(64), 0, 0, d_in, d_out, 501);
    hipMemcpy(ra, d_out, 48, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(chainF, dim3(1), dim3(64), 0, 0, d_in, d_out, 501);
    hipMemcpy(rf, d_out, 96, hipMemcpyDeviceToHost);
    bool ok = true;
    for (int i = 0; i < 6; i++) ok &= (ra[i] == rf[i]);
    printf("F(first) == A: %s\n", ok ? "YES" : "NO");
  }
  // verify D == two serial A chains
  {
    uint64_t rd[12], rs[6];
    hipLaunchKernelGGL(chainD, dim3(1), dim3(64), 0, 0, d_in, d_out, 500);
    hipMemcpy(rd, d_out, 96, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(chain<0>, dim3(1), dim3(64), 0, 0, d_in, d_out, 500);
    hipMemcpy(rs, d_out, 48, hipMemcpyDeviceToHost);
    bool ok = true;
    for (int i = 0; i < 6; i++) ok &= (rd[i] == rs[i]);
    printf("D(first chain) == A: %s\n", ok ? "YES" : "NO");
  }
  auto bench = [&](const char *name, void (*kern)(const uint64_t *, uint64_t *, int), double per) {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
      hipEventRecord(e0);
      hipLaunchKernelGGL(kern, dim3(1024), dim3(64), 0, 0, d_in, d_out, ITERS);
      hipEventRecord(e1);
      hipError_t err = hipEventSynchronize(e1);
      float ms; hipEventElapsedTime(&ms, e0, e1);
      printf("%s rep %d: %.3f ms (%s) -> %.0f cyc/op, %.0f per fp_mul-equiv\n",
             name, rep, ms, hipGetErrorString(err), ms * 1e-3 * 2.4e9 / ITERS,
             ms * 1e-3 * 2.4e9 / ITERS / per);
    }
  };
  bench("variant G0 (fp2 via 12x32 CIOS)", chainG0, 3.0);
  bench("variant G  (fp2 LAZY 3 wide + 2 redc)", chainG, 3.0);
  {
    uint64_t rg[12], rg0[12];
    hipLaunchKernelGGL(chainG0, dim3(1), dim3(64), 0, 0, d_in, d_out, 333);
    hipMemcpy(rg0, d_out, 96, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(chainG, dim3(1), dim3(64), 0, 0, d_in, d_out, 333);
    hipMemcpy(rg, d_out, 96, hipMemcpyDeviceToHost);
    bool ok = true;
    for (int i = 0; i < 12; i++) ok &= (rg[i] == rg0[i]);
    printf("G == G0 after 333 chained fp2 muls: %s\n", ok ? "YES" : "NO");
  }
  bench("variant H0 (square via mulE(a,a))", chainH<0>, 1.0);
  bench("variant H  (specialized SOS sqr)", chainH<1>, 1.0);
  {
    uint64_t rh[6], rh0[6];
    hipLaunchKernelGGL(chainH<0>, dim3(1), dim3(64), 0, 0, d_in, d_out, 444);
    hipMemcpy(rh0, d_out, 48, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(chainH<1>, dim3(1), dim3(64), 0, 0, d_in, d_out, 444);
    hipMemcpy(rh, d_out, 48, hipMemcpyDeviceToHost);
    bool ok = true;
    for (int i = 0; i < 6; i++) ok &= (rh[i] == rh0[i]);
    printf("H == H0 after 444 chained squarings: %s\n", ok ? "YES" : "NO");
  }
  // cross-check results equal
  uint64_t ra[6], rb[6];
  hipLaunchKernelGGL(chain<0>, dim3(1), dim3(64), 0, 0, d_in, d_out, 1000);
  hipMemcpy(ra, d_out, 48, hipMemcpyDeviceToHost);
  hipLaunchKernelGGL(chain<1>, dim3(1), dim3(64), 0, 0, d_in, d_out, 1000);
  hipMemcpy(rb, d_out, 48, hipMemcpyDeviceToHost);
  bool ok = true;
  for (int i = 0; i < 6; i++) ok &= (ra[i] == rb[i]);
  printf("A == B after 1000 chained muls: %s\n", ok ? "YES" : "NO");
  return 0;
}
