// Device SHA-256 (FIPS 180-4) for gfx950 — the primitive under both hot
// paths (merkleize node hashes; expand_message_xmd inside hash-to-curve).
// Replaces, on the GPU side, the role of the reference's external
// ethereum_hashing crate. Two-to-one node hashes use a
// compile-time-folded second compression (the padding block of a 64-byte
// message has a constant schedule).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

namespace m3x {

__device__ __constant__ static const uint32_t SHA_K[64] = {
    0x428a2f98u, 0x71374491u, 0xb5c0fbcfu, 0xe9b5dba5u, 0x3956c25bu,
    0x59f111f1u, 0x923f82a4u, 0xab1c5ed5u, 0xd807aa98u, 0x12835b01u,
    0x243185beu, 0x550c7dc3u, 0x72be5d74u, 0x80deb1feu, 0x9bdc06a7u,
    0xc19bf174u, 0xe49b69c1u, 0xefbe4786u, 0x0fc19dc6u, 0x240ca1ccu,
    0x2de92c6fu, 0x4a7484aau, 0x5cb0a9dcu, 0x76f988dau, 0x983e5152u,
    0xa831c66du, 0xb00327c8u, 0xbf597fc7u, 0xc6e00bf3u, 0xd5a79147u,
    0x06ca6351u, 0x14292967u, 0x27b70a85u, 0x2e1b2138u, 0x4d2c6dfcu,
    0x53380d13u, 0x650a7354u, 0x766a0abbu, 0x81c2c92eu, 0x92722c85u,
    0xa2bfe8a1u, 0xa81a664bu, 0xc24b8b70u, 0xc76c51a3u, 0xd192e819u,
    0xd6990624u, 0xf40e3585u, 0x106aa070u, 0x19a4c116u, 0x1e376c08u,
    0x2748774cu, 0x34b0bcb5u, 0x391c0cb3u, 0x4ed8aa4au, 0x5b9cca4fu,
    0x682e6ff3u, 0x748f82eeu, 0x78a5636fu, 0x84c87814u, 0x8cc70208u,
    0x90befffau, 0xa4506cebu, 0xbef9a3f7u, 0xc67178f2u};

// K[i] + W[i] of the (constant) padding block of a 64-byte message
__device__ __constant__ static const uint32_t SHA_PKW[64] = {
    0xc28a2f98u, 0x71374491u, 0xb5c0fbcfu, 0xe9b5dba5u, 0x3956c25bu,
    0x59f111f1u, 0x923f82a4u, 0xab1c5ed5u, 0xd807aa98u, 0x12835b01u,
    0x243185beu, 0x550c7dc3u, 0x72be5d74u, 0x80deb1feu, 0x9bdc06a7u,
    0xc19bf374u, 0x649b69c1u, 0xf0fe4786u, 0x0fe1edc6u, 0x240cf254u,
    0x4fe9346fu, 0x6cc984beu, 0x61b9411eu, 0x16f988fau, 0xf2c65152u,
    0xa88e5a6du, 0xb019fc65u, 0xb9d99ec7u, 0x9a1231c3u, 0xe70eeaa0u,
    0xfdb1232bu, 0xc7353eb0u, 0x3069bad5u, 0xcb976d5fu, 0x5a0f118fu,
    0xdc1eeefdu, 0x0a35b689u, 0xde0b7a04u, 0x58f4ca9du, 0xe15d5b16u,
    0x007f3e86u, 0x37088980u, 0xa507ea32u, 0x6fab9537u, 0x17406110u,
    0x0d8cd6f1u, 0xcdaa3b6du, 0xc0bbbe37u, 0x83613bdau, 0xdb48a363u,
    0x0b02e931u, 0x6fd15ca7u, 0x521afacau, 0x31338431u, 0x6ed41a95u,
    0x6d437890u, 0xc39c91f2u, 0x9eccabbdu, 0xb5c9a0e6u, 0x532fb63cu,
    0xd2c741c6u, 0x07237ea3u, 0xa4954b68u, 0x4c191d76u};

__device__ __forceinline__ uint32_t rotr32(uint32_t x, int n) {
  return __builtin_amdgcn_alignbit(x, x, n);
}

struct Sha256State {
  uint32_t h[8];
};

__device__ __forceinline__ void sha256_init(Sha256State &s) {
  s.h[0] = 0x6a09e667u; s.h[1] = 0xbb67ae85u; s.h[2] = 0x3c6ef372u;
  s.h[3] = 0xa54ff53au; s.h[4] = 0x510e527fu; s.h[5] = 0x9b05688cu;
  s.h[6] = 0x1f83d9abu; s.h[7] = 0x5be0cd19u;
}

#define M3X_SHA_ROUND(a, b, c, d, e, f, g, h, kw)                              \
  do {                                                                         \
    uint32_t S1 = rotr32(e, 6) ^ rotr32(e, 11) ^ rotr32(e, 25);                \
    uint32_t ch = (e & f) ^ (~e & g);                                          \
    uint32_t t1 = h + S1 + ch + (kw);                                          \
    uint32_t S0 = rotr32(a, 2) ^ rotr32(a, 13) ^ rotr32(a, 22);                \
    uint32_t mj = (a & b) ^ (a & c) ^ (b & c);                                 \
    h = g; g = f; f = e; e = d + t1;                                           \
    d = c; c = b; b = a; a = t1 + S0 + mj;                                     \
  } while (0)

// one compression of a 16-word big-endian block
__device__ __forceinline__ void sha256_compress(Sha256State &s,
                                                const uint32_t w_in[16]) {
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 16; i++) w[i] = w_in[i];
  uint32_t a = s.h[0], b = s.h[1], c = s.h[2], d = s.h[3], e = s.h[4],
           f = s.h[5], g = s.h[6], h = s.h[7];
#pragma unroll
  for (int i = 0; i < 16; i++) M3X_SHA_ROUND(a, b, c, d, e, f, g, h, SHA_K[i] + w[i]);
#pragma unroll
  for (int i = 16; i < 64; i++) {
    uint32_t w15 = w[(i - 15) & 15], w2 = w[(i - 2) & 15];
    uint32_t s0 = rotr32(w15, 7) ^ rotr32(w15, 18) ^ (w15 >> 3);
    uint32_t s1 = rotr32(w2, 17) ^ rotr32(w2, 19) ^ (w2 >> 10);
    uint32_t wi = w[i & 15] + s0 + w[(i - 7) & 15] + s1;
    w[i & 15] = wi;
    M3X_SHA_ROUND(a, b, c, d, e, f, g, h, SHA_K[i] + wi);
  }
  s.h[0] += a; s.h[1] += b; s.h[2] += c; s.h[3] += d;
  s.h[4] += e; s.h[5] += f; s.h[6] += g; s.h[7] += h;
}

// compression of the constant padding block of a 64-byte message
__device__ __forceinline__ void sha256_compress_pad64(Sha256State &s) {
  uint32_t a = s.h[0], b = s.h[1], c = s.h[2], d = s.h[3], e = s.h[4],
           f = s.h[5], g = s.h[6], h = s.h[7];
#pragma unroll
  for (int i = 0; i < 64; i++) M3X_SHA_ROUND(a, b, c, d, e, f, g, h, SHA_PKW[i]);
  s.h[0] += a; s.h[1] += b; s.h[2] += c; s.h[3] += d;
  s.h[4] += e; s.h[5] += f; s.h[6] += g; s.h[7] += h;
}

// two-to-one node hash: out = SHA256(l[8 words BE] || r[8 words BE]).
// words are already big-endian interpreted (byteswapped at load).
__device__ __forceinline__ void sha256_node(const uint32_t l[8],
                                            const uint32_t r[8],
                                            uint32_t out[8]) {
  Sha256State s;
  sha256_init(s);
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 8; i++) w[i] = l[i];
#pragma unroll
  for (int i = 0; i < 8; i++) w[8 + i] = r[i];
  sha256_compress(s, w);
  sha256_compress_pad64(s);
#pragma unroll
  for (int i = 0; i < 8; i++) out[i] = s.h[i];
}

// full-message SHA256 over an arbitrary byte buffer held in registers/LDS
// (used for expand_message_xmd; len <= 255 here)
__device__ inline void sha256_bytes(const uint8_t *data, uint32_t len,
                                    uint8_t out[32]) {
  Sha256State s;
  sha256_init(s);
  uint32_t off = 0;
  uint32_t w[16];
  while (len - off >= 64) {
#pragma unroll
    for (int i = 0; i < 16; i++) {
      const uint8_t *p = data + off + 4 * i;
      w[i] = ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
             ((uint32_t)p[2] << 8) | p[3];
    }
    sha256_compress(s, w);
    off += 64;
  }
  uint8_t tail[128];
  uint32_t rem = len - off;
  for (uint32_t i = 0; i < 128; i++) tail[i] = 0;
  for (uint32_t i = 0; i < rem; i++) tail[i] = data[off + i];
  tail[rem] = 0x80;
  uint32_t tlen = (rem + 9 <= 64) ? 64 : 128;
  uint64_t bits = (uint64_t)len * 8;
  for (int i = 0; i < 8; i++) tail[tlen - 1 - i] = (uint8_t)(bits >> (8 * i));
  for (uint32_t b = 0; b < tlen; b += 64) {
#pragma unroll
    for (int i = 0; i < 16; i++) {
      const uint8_t *p = tail + b + 4 * i;
      w[i] = ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
             ((uint32_t)p[2] << 8) | p[3];
    }
    sha256_compress(s, w);
  }
#pragma unroll
  for (int i = 0; i < 8; i++) {
    out[4 * i] = (uint8_t)(s.h[i] >> 24);
    out[4 * i + 1] = (uint8_t)(s.h[i] >> 16);
    out[4 * i + 2] = (uint8_t)(s.h[i] >> 8);
    out[4 * i + 3] = (uint8_t)s.h[i];
  }
}

} // namespace m3x
