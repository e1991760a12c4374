/* swap-or-not shuffle — CPU oracle restatement of
 * /root/reference/consensus/swap_or_not_shuffle/src/shuffle_list.rs
 * (the third SHA256 consumer on the block-import path;
 * 90 rounds, chain_spec.rs:632). TEST INFRASTRUCTURE ONLY.
 *
 * Returns 0 on success, -1 on the reference's None conditions
 * (empty list, list_size > 2^24, rounds == 0). indices are u32. */
#include "oracle.h"
#include <string.h>

int m3x_oracle_shuffle_list(uint32_t *input, uint64_t list_size,
                            uint8_t rounds, const uint8_t seed[32],
                            int forwards) {
  if (list_size == 0 || list_size > (1ull << 24) || rounds == 0) return -1;
  uint8_t buf[37]; /* seed(32) | round(1) | position window(4) */
  memcpy(buf, seed, 32);
  int r = forwards ? 0 : rounds - 1;
  for (;;) {
    buf[32] = (uint8_t)r;
    uint8_t digest[32];
    m3x_oracle_sha256(buf, 33, digest);
    uint64_t raw_pivot = 0;
    for (int b = 7; b >= 0; b--) raw_pivot = (raw_pivot << 8) | digest[b];
    uint64_t pivot = raw_pivot % list_size;

    uint64_t mirror = (pivot + 1) >> 1;
    uint32_t pos = (uint32_t)(pivot >> 8);
    memcpy(buf + 33, &pos, 4); /* LE */
    uint8_t source[32];
    m3x_oracle_sha256(buf, 37, source);
    uint8_t byte_v = source[(pivot & 0xff) >> 3];
    for (uint64_t i = 0; i < mirror; i++) {
      uint64_t j = pivot - i;
      if ((j & 0xff) == 0xff) {
        pos = (uint32_t)(j >> 8);
        memcpy(buf + 33, &pos, 4);
        m3x_oracle_sha256(buf, 37, source);
      }
      if ((j & 0x07) == 0x07) byte_v = source[(j & 0xff) >> 3];
      if ((byte_v >> (j & 0x07)) & 1) {
        uint32_t t = input[i];
        input[i] = input[j];
        input[j] = t;
      }
    }

    mirror = (pivot + list_size + 1) >> 1;
    uint64_t end = list_size - 1;
    pos = (uint32_t)(end >> 8);
    memcpy(buf + 33, &pos, 4);
    m3x_oracle_sha256(buf, 37, source);
    byte_v = source[(end & 0xff) >> 3];
    uint64_t loop_iter = 0;
    for (uint64_t i = pivot + 1; i < mirror; i++, loop_iter++) {
      uint64_t j = end - loop_iter;
      if ((j & 0xff) == 0xff) {
        pos = (uint32_t)(j >> 8);
        memcpy(buf + 33, &pos, 4);
        m3x_oracle_sha256(buf, 37, source);
      }
      if ((j & 0x07) == 0x07) byte_v = source[(j & 0xff) >> 3];
      if ((byte_v >> (j & 0x07)) & 1) {
        uint32_t t = input[i];
        input[i] = input[j];
        input[j] = t;
      }
    }

    if (forwards) {
      if (++r == rounds) break;
    } else {
      if (r == 0) break;
      r--;
    }
  }
  return 0;
}
