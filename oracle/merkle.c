/* SSZ merkleization — independent restatement for the oracle.
 *
 * Semantics restated from:
 *  - right-sparse tree + zero-subtree ladder:
 *    /root/reference/consensus/merkle_proof/src/lib.rs:9-14,68-100
 *  - mix_in_length: state_processing/src/common/deposit_data_tree.rs:26-38
 *  - Validator field layout + tree_hash derive:
 *    consensus/types/src/validator.rs:25-35 (8 field roots, LE basic packing)
 *  - registry limit 2^40: consensus/types/src/eth_spec.rs:404
 * Pinned against an independent hashlib-based recomputation in
 * tests/test_oracle_merkle.py. */
#include "oracle.h"
#include <stdlib.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

static uint8_t ZEROS[65][32];
static int zeros_init_done = 0;

static void zeros_init(void) {
  if (zeros_init_done)
    return;
  memset(ZEROS[0], 0, 32);
  for (int i = 1; i <= 64; i++)
    m3x_oracle_hash64(ZEROS[i - 1], ZEROS[i - 1], ZEROS[i]);
  zeros_init_done = 1;
}

void m3x_oracle_zero_hash(uint32_t depth, uint8_t out[32]) {
  zeros_init();
  memcpy(out, ZEROS[depth], 32);
}

void m3x_oracle_merkleize(const uint8_t *chunks, uint64_t n, uint32_t depth,
                          uint8_t out[32]) {
  zeros_init();
  if (n == 0) {
    memcpy(out, ZEROS[depth], 32);
    return;
  }
  if (depth == 0) { /* n == 1 */
    memcpy(out, chunks, 32);
    return;
  }
  /* ping-pong buffers: never reduce in place (the leaf level reads `chunks`,
   * later levels alternate between the two scratch buffers). */
  uint8_t *buf[2];
  buf[0] = malloc((size_t)((n + 1) / 2) * 32);
  buf[1] = malloc((size_t)((n + 3) / 4) * 32);
  uint64_t m = n;
  const uint8_t *src = chunks;
  int cur = 0;
  for (uint32_t level = 0; level < depth; level++) {
    uint64_t next = (m + 1) / 2;
    uint8_t *dst = buf[cur];
#ifdef _OPENMP
#pragma omp parallel for schedule(static) if (next > 1024)
#endif
    for (uint64_t i = 0; i < next; i++) {
      const uint8_t *l = src + 64 * i;
      const uint8_t *r = (2 * i + 1 < m) ? src + 64 * i + 32 : ZEROS[level];
      m3x_oracle_hash64(l, r, dst + 32 * i);
    }
    m = next;
    src = dst;
    cur ^= 1;
  }
  memcpy(out, src, 32);
  free(buf[0]);
  free(buf[1]);
}

void m3x_oracle_mix_in_length(const uint8_t root[32], uint64_t length,
                              uint8_t out[32]) {
  uint8_t len_chunk[32];
  memset(len_chunk, 0, 32);
  for (int i = 0; i < 8; i++)
    len_chunk[i] = (uint8_t)(length >> (8 * i));
  m3x_oracle_hash64(root, len_chunk, out);
}

static uint32_t ceil_log2(uint64_t x) {
  uint32_t d = 0;
  uint64_t c = 1;
  while (c < x) {
    c <<= 1;
    d++;
  }
  return d;
}

void m3x_oracle_validator_leaf(const uint8_t ssz[121], uint8_t out[32]) {
  /* SSZ fixed layout: pubkey[0..48] wc[48..80] eff_bal[80..88] slashed[88]
   * activation_eligibility[89..97] activation[97..105] exit[105..113]
   * withdrawable[113..121] */
  uint8_t c[8][32];
  memset(c, 0, sizeof(c));
  /* field 0: pubkey (Vector[u8,48]) root = merkleize 2 chunks */
  {
    uint8_t pk1[32];
    memset(pk1, 0, 32);
    memcpy(pk1, ssz + 32, 16);
    m3x_oracle_hash64(ssz, pk1, c[0]);
  }
  memcpy(c[1], ssz + 48, 32); /* withdrawal_credentials */
  memcpy(c[2], ssz + 80, 8);  /* effective_balance LE */
  c[3][0] = ssz[88];          /* slashed */
  memcpy(c[4], ssz + 89, 8);
  memcpy(c[5], ssz + 97, 8);
  memcpy(c[6], ssz + 105, 8);
  memcpy(c[7], ssz + 113, 8);
  /* fixed 8-leaf tree, unrolled (no per-validator malloc — hot path) */
  uint8_t l1[4][32], l2[2][32];
  m3x_oracle_hash64(c[0], c[1], l1[0]);
  m3x_oracle_hash64(c[2], c[3], l1[1]);
  m3x_oracle_hash64(c[4], c[5], l1[2]);
  m3x_oracle_hash64(c[6], c[7], l1[3]);
  m3x_oracle_hash64(l1[0], l1[1], l2[0]);
  m3x_oracle_hash64(l1[2], l1[3], l2[1]);
  m3x_oracle_hash64(l2[0], l2[1], out);
}

void m3x_oracle_validator_registry_root(const uint8_t *ssz, uint64_t n,
                                        uint8_t out[32]) {
  zeros_init();
  uint8_t *leaves = malloc((size_t)(n ? n : 1) * 32);
#ifdef _OPENMP
#pragma omp parallel for schedule(static) if (n > 256)
#endif
  for (uint64_t i = 0; i < n; i++)
    m3x_oracle_validator_leaf(ssz + 121 * i, leaves + 32 * i);
  uint8_t root[32];
  m3x_oracle_merkleize(leaves, n, 40, root); /* limit 2^40, eth_spec.rs:404 */
  m3x_oracle_mix_in_length(root, n, out);
  free(leaves);
}

/* Pack basic elements into 32B chunks (SSZ little-endian packing) and
 * merkleize to the chunk limit implied by limit_elems. */
static void packed_root(const uint8_t *data, uint64_t n_elems,
                        uint32_t elem_size, uint64_t limit_elems,
                        uint8_t out[32]) {
  uint64_t nbytes = n_elems * elem_size;
  uint64_t n_chunks = (nbytes + 31) / 32;
  uint64_t limit_bytes = limit_elems * elem_size;
  uint64_t limit_chunks = (limit_bytes + 31) / 32;
  uint32_t depth = ceil_log2(limit_chunks ? limit_chunks : 1);
  uint8_t *chunks = malloc((size_t)(n_chunks ? n_chunks : 1) * 32);
  if (n_chunks) {
    memset(chunks + (n_chunks - 1) * 32, 0, 32); /* zero-pad the tail chunk */
    memcpy(chunks, data, nbytes);
  }
  m3x_oracle_merkleize(chunks, n_chunks, depth, out);
  free(chunks);
}

void m3x_oracle_basic_list_root(const uint8_t *data, uint64_t n_elems,
                                uint32_t elem_size, uint64_t limit_elems,
                                uint8_t out[32]) {
  uint8_t root[32];
  packed_root(data, n_elems, elem_size, limit_elems, root);
  m3x_oracle_mix_in_length(root, n_elems, out);
}

void m3x_oracle_basic_vector_root(const uint8_t *data, uint64_t n_elems,
                                  uint32_t elem_size, uint8_t out[32]) {
  packed_root(data, n_elems, elem_size, n_elems, out);
}

void m3x_oracle_root_vector_root(const uint8_t *roots, uint64_t n,
                                 uint8_t out[32]) {
  m3x_oracle_merkleize(roots, n, ceil_log2(n ? n : 1), out);
}

void m3x_oracle_root_list_root(const uint8_t *roots, uint64_t n,
                               uint64_t limit, uint8_t out[32]) {
  uint8_t root[32];
  m3x_oracle_merkleize(roots, n, ceil_log2(limit ? limit : 1), root);
  m3x_oracle_mix_in_length(root, n, out);
}
