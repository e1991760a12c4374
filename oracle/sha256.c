/* FIPS 180-4 SHA-256 — independent restatement for the oracle.
 * Replaces, at the oracle level, the primitive surface of the external
 * ethereum_hashing 0.6.0 crate (Cargo.lock:2673; call sites listed in
 * SURVEY.md §2) whose `hash`/`hash32_concat` the reference uses for all
 * merkleization. Pinned by NIST KATs + hashlib golden vectors in
 * tests/test_oracle_sha256.py. */
#include "oracle.h"
#include <string.h>

static const uint32_t K[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

static const uint32_t IV[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                               0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};

#define ROTR(x, n) (((x) >> (n)) | ((x) << (32 - (n))))

static void compress(uint32_t st[8], const uint8_t block[64]) {
  uint32_t w[64];
  for (int i = 0; i < 16; i++)
    w[i] = ((uint32_t)block[4 * i] << 24) | ((uint32_t)block[4 * i + 1] << 16) |
           ((uint32_t)block[4 * i + 2] << 8) | (uint32_t)block[4 * i + 3];
  for (int i = 16; i < 64; i++) {
    uint32_t s0 = ROTR(w[i - 15], 7) ^ ROTR(w[i - 15], 18) ^ (w[i - 15] >> 3);
    uint32_t s1 = ROTR(w[i - 2], 17) ^ ROTR(w[i - 2], 19) ^ (w[i - 2] >> 10);
    w[i] = w[i - 16] + s0 + w[i - 7] + s1;
  }
  uint32_t a = st[0], b = st[1], c = st[2], d = st[3], e = st[4], f = st[5],
           g = st[6], h = st[7];
  for (int i = 0; i < 64; i++) {
    uint32_t S1 = ROTR(e, 6) ^ ROTR(e, 11) ^ ROTR(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = h + S1 + ch + K[i] + w[i];
    uint32_t S0 = ROTR(a, 2) ^ ROTR(a, 13) ^ ROTR(a, 22);
    uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + maj;
    h = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  st[0] += a; st[1] += b; st[2] += c; st[3] += d;
  st[4] += e; st[5] += f; st[6] += g; st[7] += h;
}

static void compress_blocks(uint32_t st[8], const uint8_t *data, size_t nblk);

void m3x_oracle_sha256(const uint8_t *data, size_t len, uint8_t out[32]) {
  uint32_t st[8];
  memcpy(st, IV, sizeof(IV));
  size_t off = 0;
  size_t nfull = len / 64;
  if (nfull) {
    compress_blocks(st, data, nfull);
    off = nfull * 64;
  }
  uint8_t tail[128];
  size_t rem = len - off;
  memset(tail, 0, sizeof(tail));
  memcpy(tail, data + off, rem);
  tail[rem] = 0x80;
  size_t tlen = (rem + 1 + 8 <= 64) ? 64 : 128;
  uint64_t bits = (uint64_t)len * 8;
  for (int i = 0; i < 8; i++)
    tail[tlen - 1 - i] = (uint8_t)(bits >> (8 * i));
  compress_blocks(st, tail, tlen / 64);
  for (int i = 0; i < 8; i++) {
    out[4 * i] = (uint8_t)(st[i] >> 24);
    out[4 * i + 1] = (uint8_t)(st[i] >> 16);
    out[4 * i + 2] = (uint8_t)(st[i] >> 8);
    out[4 * i + 3] = (uint8_t)st[i];
  }
}

void m3x_oracle_hash64(const uint8_t left[32], const uint8_t right[32],
                       uint8_t out[32]) {
  /* two-to-one: block 1 = left||right, block 2 = the constant padding
     block (0x80, zeros, bit length 512) */
  static const uint8_t PAD512[64] = {
      0x80, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
      0,    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
      0,    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
      0,    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0x02, 0x00};
  uint32_t st[8];
  memcpy(st, IV, sizeof(IV));
  uint8_t buf[128];
  memcpy(buf, left, 32);
  memcpy(buf + 32, right, 32);
  memcpy(buf + 64, PAD512, 64);
  compress_blocks(st, buf, 2);
  for (int i = 0; i < 8; i++) {
    out[4 * i] = (uint8_t)(st[i] >> 24);
    out[4 * i + 1] = (uint8_t)(st[i] >> 16);
    out[4 * i + 2] = (uint8_t)(st[i] >> 8);
    out[4 * i + 3] = (uint8_t)st[i];
  }
}

/* ---------------- SHA-NI (x86 SHA extensions) fast path ----------------
 * Runtime-dispatched: engaged when the CPU reports SHA extensions (this
 * survey container's Xeon and EPYC GPU-box hosts both have sha_ni).
 * Same FIPS 180-4 function — pinned by the same NIST KATs + hashlib
 * golden vectors as the portable path (tests run both). */
#if defined(__x86_64__)
#include <immintrin.h>

__attribute__((target("sha,sse4.1,ssse3")))
static void compress_ni(uint32_t st[8], const uint8_t *data, size_t nblk) {
  const __m128i MASK =
      _mm_set_epi64x(0x0c0d0e0f08090a0bULL, 0x0405060700010203ULL);
  __m128i T0 = _mm_loadu_si128((const __m128i *)st);
  __m128i T1 = _mm_loadu_si128((const __m128i *)(st + 4));
  T0 = _mm_shuffle_epi32(T0, 0xB1);         /* CDAB */
  T1 = _mm_shuffle_epi32(T1, 0x1B);         /* EFGH */
  __m128i S0 = _mm_alignr_epi8(T0, T1, 8);  /* ABEF */
  __m128i S1 = _mm_blend_epi16(T1, T0, 0xF0); /* CDGH */
  while (nblk--) {
    const __m128i AS = S0, CS = S1;
    __m128i MSG, TMP;
    __m128i M0 = _mm_shuffle_epi8(_mm_loadu_si128((const __m128i *)(data + 0)), MASK);
    __m128i M1 = _mm_shuffle_epi8(_mm_loadu_si128((const __m128i *)(data + 16)), MASK);
    __m128i M2 = _mm_shuffle_epi8(_mm_loadu_si128((const __m128i *)(data + 32)), MASK);
    __m128i M3 = _mm_shuffle_epi8(_mm_loadu_si128((const __m128i *)(data + 48)), MASK);
#define M3X_QROUND(M, k)                                                       \
  MSG = _mm_add_epi32(M, _mm_loadu_si128((const __m128i *)&K[k]));             \
  S1 = _mm_sha256rnds2_epu32(S1, S0, MSG);                                     \
  MSG = _mm_shuffle_epi32(MSG, 0x0E);                                          \
  S0 = _mm_sha256rnds2_epu32(S0, S1, MSG);
#define M3X_SCHED(Ma, Mb, Mc, Md)                                              \
  Ma = _mm_sha256msg1_epu32(Ma, Mb);                                           \
  TMP = _mm_alignr_epi8(Md, Mc, 4);                                            \
  Ma = _mm_add_epi32(Ma, TMP);                                                 \
  Ma = _mm_sha256msg2_epu32(Ma, Md);
    M3X_QROUND(M0, 0)
    M3X_QROUND(M1, 4)
    M3X_QROUND(M2, 8)
    M3X_QROUND(M3, 12)
    M3X_SCHED(M0, M1, M2, M3) M3X_QROUND(M0, 16)
    M3X_SCHED(M1, M2, M3, M0) M3X_QROUND(M1, 20)
    M3X_SCHED(M2, M3, M0, M1) M3X_QROUND(M2, 24)
    M3X_SCHED(M3, M0, M1, M2) M3X_QROUND(M3, 28)
    M3X_SCHED(M0, M1, M2, M3) M3X_QROUND(M0, 32)
    M3X_SCHED(M1, M2, M3, M0) M3X_QROUND(M1, 36)
    M3X_SCHED(M2, M3, M0, M1) M3X_QROUND(M2, 40)
    M3X_SCHED(M3, M0, M1, M2) M3X_QROUND(M3, 44)
    M3X_SCHED(M0, M1, M2, M3) M3X_QROUND(M0, 48)
    M3X_SCHED(M1, M2, M3, M0) M3X_QROUND(M1, 52)
    M3X_SCHED(M2, M3, M0, M1) M3X_QROUND(M2, 56)
    M3X_SCHED(M3, M0, M1, M2) M3X_QROUND(M3, 60)
#undef M3X_QROUND
#undef M3X_SCHED
    S0 = _mm_add_epi32(S0, AS);
    S1 = _mm_add_epi32(S1, CS);
    data += 64;
  }
  T0 = _mm_shuffle_epi32(S0, 0x1B);           /* FEBA */
  T1 = _mm_shuffle_epi32(S1, 0xB1);           /* DCHG */
  S0 = _mm_blend_epi16(T0, T1, 0xF0);         /* DCBA */
  S1 = _mm_alignr_epi8(T1, T0, 8);            /* HGFE */
  _mm_storeu_si128((__m128i *)st, S0);
  _mm_storeu_si128((__m128i *)(st + 4), S1);
}

static int sha_ni_ok = -1;
static int have_sha_ni(void) {
  if (sha_ni_ok < 0) sha_ni_ok = __builtin_cpu_supports("sha") ? 1 : 0;
  return sha_ni_ok;
}
#else
static int have_sha_ni(void) { return 0; }
#endif

int m3x_oracle_have_sha_ni(void) { return have_sha_ni(); }

static void compress_blocks(uint32_t st[8], const uint8_t *data, size_t nblk) {
#if defined(__x86_64__)
  if (have_sha_ni()) {
    compress_ni(st, data, nblk);
    return;
  }
#endif
  while (nblk--) {
    compress(st, data);
    data += 64;
  }
}
