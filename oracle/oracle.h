/* oracle/ — CPU restatement of the reference's hot-path algorithms.
 *
 * TEST INFRASTRUCTURE ONLY. Per DESIGN.md, only tests/, __graft_entry__'s
 * smoke checker and bench.py's cpu_baseline leg may call this library; the
 * product path (lighthouse_amd + libm3x_consensus.so) never does.
 *
 * Parity pinning: SHA256 against FIPS 180-4 KATs and Python hashlib; SSZ
 * merkleize against the semantics of
 * /root/reference/consensus/merkle_proof/src/lib.rs:9-14,68-100 (zero-hash
 * ladder, right-sparse create), deposit_data_tree.rs:26-38 (mix_in_length),
 * consensus/types/src/validator.rs:25-35 (Validator field layout, tree_hash
 * derive), eth_spec.rs:404 (2^40 registry limit). BLS: see bls12_381.h.
 */
#ifndef M3X_ORACLE_H
#define M3X_ORACLE_H
#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* FIPS 180-4 SHA-256 of an arbitrary message. */
void m3x_oracle_sha256(const uint8_t *data, size_t len, uint8_t out[32]);

/* hash32_concat: SHA256(left||right) — the merkle two-to-one node hash
 * (ethereum_hashing::hash32_concat call sites: merkle_proof/src/lib.rs:91). */
void m3x_oracle_hash64(const uint8_t left[32], const uint8_t right[32],
                       uint8_t out[32]);

/* zero-hash ladder: out = Z[depth] where Z[0]=0^32, Z[i]=H(Z[i-1],Z[i-1])
 * (merkle_proof/src/lib.rs:9-14 ZERO_NODES). depth <= 64. */
void m3x_oracle_zero_hash(uint32_t depth, uint8_t out[32]);

/* Right-sparse merkleize of n 32-byte chunks into a tree of the given depth
 * (capacity 2^depth leaves, missing leaves are zero chunks) — the iterative
 * equivalent of MerkleTree::create (merkle_proof/src/lib.rs:68-100).
 * n may be 0; requires n <= 2^depth. */
void m3x_oracle_merkleize(const uint8_t *chunks, uint64_t n, uint32_t depth,
                          uint8_t out[32]);

/* root = H(root || LE64(length) padded to 32B)
 * (deposit_data_tree.rs:26-38; tree_hash mix_in_length). */
void m3x_oracle_mix_in_length(const uint8_t root[32], uint64_t length,
                              uint8_t out[32]);

/* hash_tree_root of one Validator from its 121-byte SSZ encoding
 * (validator.rs:25-35; fields pubkey48|wc32|eff_bal8|slashed1|4 epochs). */
void m3x_oracle_validator_leaf(const uint8_t ssz[121], uint8_t out[32]);

/* hash_tree_root of List[Validator, 2^40] (eth_spec.rs:404) from packed
 * 121-byte SSZ records: per-validator leaves, depth-40 merkleize,
 * mix_in_length(n). OpenMP-parallel. */
void m3x_oracle_validator_registry_root(const uint8_t *ssz, uint64_t n,
                                        uint8_t out[32]);

/* hash_tree_root of a List of basic elements (uintN/byte), packed 32B-chunk
 * little-endian per SSZ: merkleize(pack(data), depth=ceil_log2(chunk_limit))
 * then mix_in_length(n_elems). elem_size in bytes (1 or 8 used here). */
void m3x_oracle_basic_list_root(const uint8_t *data, uint64_t n_elems,
                                uint32_t elem_size, uint64_t limit_elems,
                                uint8_t out[32]);

/* hash_tree_root of a Vector of basic elements (no length mix; capacity =
 * exactly n_elems = the type's length). */
void m3x_oracle_basic_vector_root(const uint8_t *data, uint64_t n_elems,
                                  uint32_t elem_size, uint8_t out[32]);

/* hash_tree_root of Vector[Hash256, n] (chunks are the elements). */
void m3x_oracle_root_vector_root(const uint8_t *roots, uint64_t n,
                                 uint8_t out[32]);

/* hash_tree_root of List[Hash256, limit] */
void m3x_oracle_root_list_root(const uint8_t *roots, uint64_t n,
                               uint64_t limit, uint8_t out[32]);

/* swap-or-not shuffle (shuffle_list.rs restatement; §8f.1). In-place on
 * u32 indices; 0 ok, -1 on the reference's None conditions. */
int m3x_oracle_shuffle_list(uint32_t *input, uint64_t list_size,
                            uint8_t rounds, const uint8_t seed[32],
                            int forwards);

#ifdef __cplusplus
}
#endif
#endif
