/* m3x_consensus — C-ABI of the MI355X-native consensus hot-path library.
 *
 * This is the FFI boundary a Lighthouse-class host binds against (see
 * INTEGRATION.md for the Rust-side binding). Each entry point cites the
 * reference interface it replaces:
 *
 *  - m3x_bls_verify_sets        <- crypto/bls/src/impls/blst.rs:37-119
 *                                  (bls::verify_signature_sets; the batch
 *                                  equation with host-drawn 64-bit r_i)
 *  - m3x_bls_pk_decompress      <- blst.rs:130-153 key_validate, as used by
 *                                  validator_pubkey_cache.rs:20-46
 *  - m3x_merkleize_validators   <- BeaconState::update_validators_tree_hash_cache
 *                                  (consensus/types/src/beacon_state.rs:2043;
 *                                  List[Validator, 2^40], eth_spec.rs:404)
 *  - m3x_merkleize_chunks       <- tree_hash merkle_root + mix_in_length
 *                                  (merkle_proof/src/lib.rs:68-100,
 *                                  deposit_data_tree.rs:26-38)
 *
 * Conventions: plain pointers + sizes, caller-owned host buffers,
 * synchronous calls (internally pipelined H2D/compute), errors as negative
 * codes, verification failure is 0 (false) — never an error (the
 * crypto/bls lib.rs:49-62 contract). The *_dev variants take device
 * pointers obtained from m3x_dev_alloc/m3x_h2d so callers can keep inputs
 * resident in HBM across calls. All functions are thread-safe w.r.t.
 * distinct contexts; one context serializes on its own stream.
 */
#ifndef M3X_CONSENSUS_H
#define M3X_CONSENSUS_H
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct m3x_ctx m3x_ctx;

/* error codes */
#define M3X_OK 0
#define M3X_ERR_HIP -1      /* HIP runtime failure */
#define M3X_ERR_ARG -2      /* bad argument */
#define M3X_ERR_NOMEM -3

int32_t m3x_ctx_create(m3x_ctx **out, int32_t device);
void m3x_ctx_destroy(m3x_ctx *ctx);
/* version/capability probe (also serves as a loadability check) */
int32_t m3x_abi_version(void);

/* per-kernel HIP-event timing (for benchmarks/roofline): enable resets the
 * accumulators; kernel ids are the M3X_K_* slots (0 leaves, 1 reduce,
 * 2 finalize, 3 bls_prepare, 4 bls_h2c, 5 bls_miller, 6 bls_reduce,
 * 7 bls_finish). */
int32_t m3x_timing_enable(m3x_ctx *ctx, int32_t on);
int32_t m3x_kernel_ms(m3x_ctx *ctx, int32_t kernel_id, double *ms,
                      uint64_t *launches);

/* carry a 32B subtree root from `from_level` to `to_depth` against the
 * zero ladder, then optionally mix_in_length — the multi-GPU cap-finishing
 * primitive paired with m3x_validator_subtree_root_dev. */
int32_t m3x_finalize_root(m3x_ctx *ctx, const uint8_t node[32],
                          uint32_t from_level, uint32_t to_depth,
                          int64_t mix_len, uint8_t out_root[32]);

/* ---- device buffer management (for *_dev calls / benchmarks) ---- */
int32_t m3x_dev_alloc(m3x_ctx *ctx, uint64_t bytes, void **dev_ptr);
int32_t m3x_dev_free(m3x_ctx *ctx, void *dev_ptr);
int32_t m3x_h2d(m3x_ctx *ctx, void *dst_dev, const void *src_host,
                uint64_t bytes);
int32_t m3x_d2h(m3x_ctx *ctx, void *dst_host, const void *src_dev,
                uint64_t bytes);

/* ---- SHA256 SSZ merkleize (hot path #2) ---- */

/* hash_tree_root of List[Validator, 2^40] from packed 121-byte SSZ records
 * (validator.rs:25-35 layout), including the mix_in_length. */
int32_t m3x_merkleize_validators(m3x_ctx *ctx, const uint8_t *ssz, uint64_t n,
                                 uint8_t out_root[32]);
int32_t m3x_merkleize_validators_dev(m3x_ctx *ctx, const void *ssz_dev,
                                     uint64_t n, uint8_t out_root[32]);

/* merkleize n_chunks 32-byte chunks into a depth-`depth` tree (capacity
 * 2^depth, zero-ladder padded); if mix_len >= 0, mix_in_length(mix_len).
 * Caller zero-pads the tail chunk. */
int32_t m3x_merkleize_chunks(m3x_ctx *ctx, const uint8_t *chunks,
                             uint64_t n_chunks, uint32_t depth,
                             int64_t mix_len, uint8_t out_root[32]);
int32_t m3x_merkleize_chunks_dev(m3x_ctx *ctx, const void *chunks_dev,
                                 uint64_t n_chunks, uint32_t depth,
                                 int64_t mix_len, uint8_t out_root[32]);

/* Partial subtree root over n validator records at fixed subtree depth
 * (no zero-cap to 2^40, no length mix) — the multi-GPU shard primitive:
 * rank i computes its contiguous range's depth-d subtree root, rank 0
 * finishes with m3x_merkleize_chunks over the gathered roots. */
int32_t m3x_validator_subtree_root_dev(m3x_ctx *ctx, const void *ssz_dev,
                                       uint64_t n, uint32_t depth,
                                       uint8_t out_root[32]);

/* ---- incremental registry merkleize (SURVEY §8f.3 — the milhouse-style
 * cached rehash: beacon_state.rs:1990-2021 has_pending_updates path).
 * The cache holds every tree level in HBM; an update rehashes only the
 * dirty leaves and their root paths. ---- */
typedef struct m3x_registry_cache m3x_registry_cache;

/* build the cache from n packed 121-byte records; capacity is the next
 * power of two >= max(n, 256) (appends beyond it are an error this round). */
int32_t m3x_registry_cache_create(m3x_ctx *ctx, const uint8_t *ssz,
                                  uint64_t n, m3x_registry_cache **out);
void m3x_registry_cache_destroy(m3x_registry_cache *cache);
/* current List[Validator, 2^40] root (zero-cap + mix_in_length) */
int32_t m3x_registry_cache_root(m3x_ctx *ctx, m3x_registry_cache *cache,
                                uint8_t out_root[32]);
/* apply m updated/appended records at `indices` (each < capacity;
 * n grows to max over indices+1) and return the new root */
int32_t m3x_registry_cache_update(m3x_ctx *ctx, m3x_registry_cache *cache,
                                  const uint64_t *indices,
                                  const uint8_t *recs /* m*121 */, uint64_t m,
                                  uint8_t out_root[32]);

/* Batched small-container merkleize: element i owns chunks
 * [offsets[i], offsets[i+1]) (<=32 chunks each); out_roots[i] = its
 * hash_tree_root (depth = ceil_log2(count)). One launch for the
 * BeaconState's many tiny containers. */
int32_t m3x_merkleize_batch(m3x_ctx *ctx, const uint8_t *chunks,
                            const uint32_t *offsets, uint64_t n_elems,
                            uint8_t *out_roots);

/* swap-or-not committee shuffle — restated from
 * consensus/swap_or_not_shuffle/src/shuffle_list.rs; in-place on u32
 * indices; rounds = ChainSpec::shuffle_round_count (90 mainnet,
 * chain_spec.rs:632); forwards semantics as the reference. */
int32_t m3x_shuffle_list(m3x_ctx *ctx, uint32_t *indices, uint64_t list_size,
                         uint32_t rounds, const uint8_t seed[32],
                         int32_t forwards);
int32_t m3x_shuffle_list_dev(m3x_ctx *ctx, void *indices_dev,
                             uint64_t list_size, uint32_t rounds,
                             const uint8_t seed[32], int32_t forwards);

/* ---- BLS12-381 batched signature-set verification (hot path #1) ---- */

/* key_validate a batch of compressed pubkeys: decompress + infinity reject +
 * subgroup check; uncomp[i] valid iff status[i]==0. (pubkey cache build) */
/* RFC 9380 test entries (external-vector pinning; not on the hot path):
 * the same device expand/SSWU/isogeny/cofactor code as the verify
 * pipeline, parameterized by DST so the literal RFC appendix vectors
 * (QUUX DSTs) apply. out_uniform (256B, optional NULL) exposes the
 * expand_message_xmd output for the G2 suite's len_in_bytes=256. */
int32_t m3x_bls_expand_test(m3x_ctx *ctx, const uint8_t *msg,
                            uint32_t msg_len, const uint8_t *dst,
                            uint32_t dst_len, uint32_t len_in_bytes,
                            uint8_t *out);
int32_t m3x_bls_h2c_test(m3x_ctx *ctx, const uint8_t *msg, uint32_t msg_len,
                         const uint8_t *dst, uint32_t dst_len,
                         uint8_t out_uncomp[192], uint8_t out_uniform[256]);

int32_t m3x_bls_pk_decompress(m3x_ctx *ctx, const uint8_t *comp /* n*48 */,
                              uint64_t n, uint8_t *uncomp /* n*96 */,
                              int32_t *status /* n */);

/* The blst.rs:37-119 batch check over n signature sets:
 *   msgs: n*32 (signing roots); sigs: n*96 compressed G2 (SSZ wire form;
 *   decompressed on-GPU, subgroup checks deferred to here exactly as
 *   generic_aggregate_signature.rs:161-176 + blst.rs:73-77);
 *   pks: sum(k_i)*96 uncompressed affine G1, ALREADY validated once at
 *   cache build (validator_pubkey_cache model); pk_offsets: n+1;
 *   rands: n host-drawn 64-bit nonzero scalars (blst.rs:53-68).
 * Host-side rules the caller keeps (as the reference keeps them above
 * blst): empty set list, empty (all-zero) signatures, empty key lists.
 * Returns 1 valid / 0 invalid / <0 error. */
/* Aggregate n compressed G2 signatures into one (the TAggregateSignature
 * add_assign surface, generic_aggregate_signature.rs:124-150; used by
 * attestation aggregation). Invalid encodings -> nonzero return; the sum
 * may legitimately be the point at infinity (0xc0...). */
int32_t m3x_bls_sig_aggregate(m3x_ctx *ctx, const uint8_t *sigs /* n*96 */,
                              uint64_t n, uint8_t out_sig[96]);

int32_t m3x_bls_verify_sets(m3x_ctx *ctx, const uint8_t *msgs,
                            const uint8_t *sigs, const uint8_t *pks,
                            const uint32_t *pk_offsets, const uint64_t *rands,
                            uint64_t n);
int32_t m3x_bls_verify_sets_dev(m3x_ctx *ctx, const void *msgs_dev,
                                const void *sigs_dev, const void *pks_dev,
                                const void *pk_offsets_dev,
                                const void *rands_dev, uint64_t n);

#ifdef __cplusplus
}
#endif
#endif
