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
