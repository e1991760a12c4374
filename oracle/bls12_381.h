/* BLS12-381 CPU oracle — independent C restatement of the reference's BLS
 * hot path (TEST INFRASTRUCTURE ONLY — see oracle/oracle.h header).
 *
 * Restates, file-by-file:
 *  - batch equation + rejection rules: crypto/bls/src/impls/blst.rs:37-119
 *    (DST :15, RAND_BITS=64 :16; min_pk scheme: pk in G1, sig in G2)
 *  - encodings: generic_public_key.rs:12-21 (48B comp / 96B uncomp, ZCash
 *    flag bits), generic_signature.rs:15-26 (96B comp, infinity 0xc0...)
 *  - keygen: eth2_interop_keypairs/src/lib.rs:40-55
 *  - hash-to-curve: RFC 9380 BLS12381G2_XMD:SHA-256_SSWU_RO_ (public spec)
 * Constants come from oracle/bls_consts.h, generated + numerically validated
 * by tests/golden/gen_bls_fixtures.py (interop vectors, on-curve/order
 * asserts, psi==h_eff, bilinearity). Parity pinning status: see DESIGN.md
 * §Parity strategy — beyond keygen, blst-boundary outputs are pinned by the
 * committed Python-reference fixtures (dual implementation), not EF vectors
 * (absent offline).
 */
#ifndef M3X_ORACLE_BLS_H
#define M3X_ORACLE_BLS_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* interop keygen: sk (32B big-endian) for validator index. */
void m3x_oracle_bls_keygen(uint64_t index, uint8_t sk_be[32]);

/* pk = [sk]G1; out compressed 48B. returns 0 ok, <0 bad sk. */
int m3x_oracle_bls_sk_to_pk(const uint8_t sk_be[32], uint8_t pk_comp[48]);

/* sig = [sk]hash_to_curve(msg); out compressed 96B. */
int m3x_oracle_bls_sign(const uint8_t sk_be[32], const uint8_t msg[32],
                        uint8_t sig_comp[96]);

/* key_validate (blst.rs:130-153): decompress + infinity reject + subgroup
 * check. out = 96B uncompressed affine. 0 ok, <0 invalid. */
int m3x_oracle_bls_pk_decompress(const uint8_t pk_comp[48],
                                 uint8_t pk_uncomp[96]);

/* Signature::from_bytes: decompress WITHOUT subgroup check (deferred to
 * verify, generic_aggregate_signature.rs:161-176). infinity allowed.
 * out = 192B uncompressed. 0 ok, <0 malformed. */
int m3x_oracle_bls_sig_decompress(const uint8_t sig_comp[96],
                                  uint8_t sig_uncomp[192]);

/* single-set verify: e(pk, H(m)) == e(g1, sig), with sig subgroup check.
 * pk uncompressed 96B (pre-validated), sig compressed 96B.
 * returns 1 valid / 0 invalid. */
int m3x_oracle_bls_verify(const uint8_t pk_uncomp[96], const uint8_t msg[32],
                          const uint8_t sig_comp[96]);

/* Batch verify — the blst.rs:37-119 contract. msgs n*32, sigs n*96
 * compressed, pks sum(k_i)*96 uncompressed (pre-validated), pk_offsets n+1,
 * rands n (host-drawn 64-bit nonzero). 1 valid / 0 invalid. */
int m3x_oracle_bls_verify_sets(const uint8_t *msgs, const uint8_t *sigs,
                               const uint8_t *pks, const uint32_t *pk_offsets,
                               const uint64_t *rands, uint64_t n);


/* workload-generation helpers (OpenMP batch; tests/bench only) */
int m3x_oracle_bls_keypool(uint64_t n, uint8_t *sks, uint8_t *pks_uncomp);
int m3x_oracle_bls_sign_batch(uint64_t n, const uint8_t *sks,
                              const uint8_t *msgs, uint8_t *sigs);

/* introspection for parity tests */
int m3x_oracle_bls_h2c_g2(const uint8_t msg[32], uint8_t out_uncomp[192]);
/* e(P,Q): P 96B uncomp G1, Q 192B uncomp G2; out 576B (12*48 BE,
 * coefficients c0..c5 of Fp2[w]/(w^6-xi), each c0||c1). */
int m3x_oracle_bls_pairing(const uint8_t p_uncomp[96],
                           const uint8_t q_uncomp[192], uint8_t out[576]);
/* RFC 9380 general forms (external-vector pinning; tests only) */
void m3x_oracle_expand_xmd(const uint8_t *msg, uint32_t msg_len,
                           const uint8_t *dst, uint32_t dst_len,
                           uint32_t len_in_bytes, uint8_t *out);
int m3x_oracle_h2c_g2_dst(const uint8_t *msg, uint32_t msg_len,
                          const uint8_t *dst, uint32_t dst_len,
                          uint8_t out_uncomp[192]);
int m3x_oracle_map_to_curve_g2_nococlear(const uint8_t msg[32],
                                         uint8_t out_uncomp[192]);

int m3x_oracle_g2_subgroup_check(const uint8_t uncomp[192], int use_ref);
int m3x_oracle_bls_selftest(void);

/* [scalar_be32]P for G1 (96B uncomp in/out); for tests. */
int m3x_oracle_bls_g1_mul(const uint8_t p_uncomp[96],
                          const uint8_t scalar_be[32], uint8_t out[96]);

#ifdef __cplusplus
}
#endif
#endif
