/* Public oracle BLS APIs — the blst.rs:37-119 batch-verify contract and the
 * keygen/sign/verify surface used by tests. TEST INFRASTRUCTURE ONLY
 * (see oracle/oracle.h header). */
#include "bls12_381.h"
#include "bls_internal.h"
#include "bls_consts.h"
#include "oracle.h"
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

/* ----- 256-bit big-endian scalar helpers (mod r) ----- */

static int be32_ge(const uint8_t a[32], const uint8_t b[32]) {
  for (int i = 0; i < 32; i++) {
    if (a[i] > b[i]) return 1;
    if (a[i] < b[i]) return 0;
  }
  return 1;
}

static void be32_sub(uint8_t a[32], const uint8_t b[32]) {
  int borrow = 0;
  for (int i = 31; i >= 0; i--) {
    int d = (int)a[i] - b[i] - borrow;
    borrow = d < 0;
    a[i] = (uint8_t)(d & 0xFF);
  }
}

static int be32_is_zero(const uint8_t a[32]) {
  uint8_t o = 0;
  for (int i = 0; i < 32; i++) o |= a[i];
  return o == 0;
}

void m3x_oracle_bls_keygen(uint64_t index, uint8_t sk_be[32]) {
  /* eth2_interop_keypairs/src/lib.rs:40-55: sk = LE(SHA256(LE64(index) pad
   * to 32)) mod r, output big-endian */
  bls_init();
  uint8_t pre[32] = {0};
  for (int i = 0; i < 8; i++) pre[i] = (uint8_t)(index >> (8 * i));
  uint8_t h[32];
  m3x_oracle_sha256(pre, 32, h);
  uint8_t be[32];
  for (int i = 0; i < 32; i++) be[i] = h[31 - i]; /* LE -> BE */
  while (be32_ge(be, ORDER_BE)) be32_sub(be, ORDER_BE);
  memcpy(sk_be, be, 32);
}

static int sk_valid(const uint8_t sk_be[32]) {
  return !be32_is_zero(sk_be) && !be32_ge(sk_be, ORDER_BE);
}

int m3x_oracle_bls_sk_to_pk(const uint8_t sk_be[32], uint8_t pk_comp[48]) {
  bls_init();
  if (!sk_valid(sk_be)) return -1;
  g1_jac_t j;
  g1_mul_be(&j, &G1_GEN, sk_be, 32);
  g1_aff_t a;
  g1_to_aff(&a, &j);
  g1_compress(&a, pk_comp);
  return 0;
}

int m3x_oracle_bls_sign(const uint8_t sk_be[32], const uint8_t msg[32],
                        uint8_t sig_comp[96]) {
  bls_init();
  if (!sk_valid(sk_be)) return -1;
  g2_aff_t h;
  h2c_g2(&h, msg);
  g2_jac_t j;
  g2_mul_be(&j, &h, sk_be, 32);
  g2_aff_t a;
  g2_to_aff(&a, &j);
  g2_compress(&a, sig_comp);
  return 0;
}

int m3x_oracle_bls_pk_decompress(const uint8_t pk_comp[48],
                                 uint8_t pk_uncomp[96]) {
  /* key_validate semantics (blst.rs:130-153) + infinity rejection
   * (generic_public_key.rs:86-94) */
  bls_init();
  g1_aff_t p;
  if (g1_decompress(&p, pk_comp) != 0) return -1;
  if (p.inf) return -2;
  if (!g1_on_curve(&p)) return -3;
  if (!g1_in_subgroup(&p)) return -4;
  g1_to_uncomp(&p, pk_uncomp);
  return 0;
}

int m3x_oracle_bls_sig_decompress(const uint8_t sig_comp[96],
                                  uint8_t sig_uncomp[192]) {
  /* Signature::from_bytes: no subgroup check (deferred to verify,
   * generic_aggregate_signature.rs:161-176); infinity allowed */
  bls_init();
  g2_aff_t p;
  if (g2_decompress(&p, sig_comp) != 0) return -1;
  g2_to_uncomp(&p, sig_uncomp);
  return 0;
}

int m3x_oracle_bls_verify(const uint8_t pk_uncomp[96], const uint8_t msg[32],
                          const uint8_t sig_comp[96]) {
  /* e(pk, H(m)) == e(g1, sig)  <=>  e(-pk, H(m)) * e(g1, sig) == 1,
   * with the signature subgroup-checked (blst.rs:196-200 contract) */
  bls_init();
  g1_aff_t pk;
  if (g1_from_uncomp(&pk, pk_uncomp) != 0 || pk.inf) return 0;
  g2_aff_t sig;
  if (g2_decompress(&sig, sig_comp) != 0) return 0;
  if (!g2_in_subgroup(&sig)) return 0;
  g1_aff_t npk = pk;
  fp_t ny;
  fp_sub_(&ny, &(fp_t){{0}}, &npk.y);
  npk.y = ny;
  g2_aff_t h;
  h2c_g2(&h, msg);
  fp12_t f;
  fp12_one(&f);
  miller(&f, &npk, &h);
  miller(&f, &G1_GEN, &sig);
  fp12_t e;
  final_exp3(&e, &f); /* cubed hard part: equivalent for the ==1 test */
  return fp12_is_one(&e);
}

int m3x_oracle_bls_verify_sets(const uint8_t *msgs, const uint8_t *sigs,
                               const uint8_t *pks, const uint32_t *pk_offsets,
                               const uint64_t *rands, uint64_t n) {
  /* blst.rs:37-119: per set — decompress+subgroup-check sigma, require
   * non-empty keys, aggregate pubkeys; then
   * prod_i e(r_i*PK_i, H(m_i)) * e(-g1, sum_i r_i*sigma_i) == 1 */
  bls_init();
  if (n == 0) return 0;
  int fail = 0;
  fp12_t f_total;
  fp12_one(&f_total);
  g2_jac_t sig_total;
  memset(&sig_total, 0, sizeof(sig_total));
#ifdef _OPENMP
#pragma omp parallel
#endif
  {
    fp12_t f_loc;
    fp12_one(&f_loc);
    g2_jac_t sig_loc;
    memset(&sig_loc, 0, sizeof(sig_loc));
#ifdef _OPENMP
#pragma omp for schedule(dynamic, 1)
#endif
    for (int64_t i = 0; i < (int64_t)n; i++) {
      if (__atomic_load_n(&fail, __ATOMIC_RELAXED)) continue;
      g2_aff_t sig;
      if (g2_decompress(&sig, sigs + 96 * i) != 0 || !g2_in_subgroup(&sig)) {
        __atomic_store_n(&fail, 1, __ATOMIC_RELAXED);
        continue;
      }
      uint32_t k0 = pk_offsets[i], k1 = pk_offsets[i + 1];
      if (k1 <= k0) {
        __atomic_store_n(&fail, 1, __ATOMIC_RELAXED);
        continue;
      }
      g1_jac_t apk;
      memset(&apk, 0, sizeof(apk));
      int bad = 0;
      for (uint32_t k = k0; k < k1; k++) {
        g1_aff_t pk;
        if (g1_from_uncomp(&pk, pks + 96 * (uint64_t)k) != 0) {
          bad = 1;
          break;
        }
        g1_add_aff(&apk, &apk, &pk);
      }
      if (bad || g1_jac_is_inf(&apk)) {
        __atomic_store_n(&fail, 1, __ATOMIC_RELAXED);
        continue;
      }
      /* P = [r_i] apk */
      g1_aff_t apk_a;
      g1_to_aff(&apk_a, &apk);
      uint8_t rbe[8];
      for (int b = 0; b < 8; b++) rbe[b] = (uint8_t)(rands[i] >> (56 - 8 * b));
      g1_jac_t rp;
      g1_mul_be(&rp, &apk_a, rbe, 8);
      g1_aff_t rp_a;
      g1_to_aff(&rp_a, &rp);
      g2_aff_t h;
      h2c_g2(&h, msgs + 32 * i);
      miller(&f_loc, &rp_a, &h);
      /* sig_acc += [r_i] sigma */
      if (!sig.inf) {
        g2_jac_t rs;
        g2_mul_be(&rs, &sig, rbe, 8);
        g2_addj(&sig_loc, &sig_loc, &rs);
      }
    }
#ifdef _OPENMP
#pragma omp critical
#endif
    {
      fp12_mul_(&f_total, &f_total, &f_loc);
      g2_addj(&sig_total, &sig_total, &sig_loc);
    }
  }
  if (fail) return 0;
  g1_aff_t ng1 = G1_GEN;
  fp_t zero_fp;
  memset(&zero_fp, 0, sizeof(zero_fp));
  fp_sub_(&ng1.y, &zero_fp, &G1_GEN.y);
  g2_aff_t sig_a;
  g2_to_aff(&sig_a, &sig_total);
  miller(&f_total, &ng1, &sig_a);
  fp12_t e;
  final_exp3(&e, &f_total); /* cubed hard part: equivalent for the ==1 test */
  return fp12_is_one(&e);
}

int m3x_oracle_bls_keypool(uint64_t n, uint8_t *sks /* n*32 */,
                           uint8_t *pks_uncomp /* n*96 */) {
  /* interop keypairs 0..n-1 (lib.rs:40-55), pubkeys uncompressed —
   * OpenMP-parallel workload generator for tests/bench. */
  bls_init();
  int fail = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 16)
#endif
  for (int64_t i = 0; i < (int64_t)n; i++) {
    m3x_oracle_bls_keygen((uint64_t)i, sks + 32 * i);
    g1_jac_t j;
    g1_mul_be(&j, &G1_GEN, sks + 32 * i, 32);
    g1_aff_t a;
    g1_to_aff(&a, &j);
    g1_to_uncomp(&a, pks_uncomp + 96 * i);
  }
  return fail;
}

int m3x_oracle_bls_sign_batch(uint64_t n, const uint8_t *sks /* n*32 */,
                              const uint8_t *msgs /* n*32 */,
                              uint8_t *sigs /* n*96 */) {
  bls_init();
  int fail = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 4)
#endif
  for (int64_t i = 0; i < (int64_t)n; i++) {
    if (m3x_oracle_bls_sign(sks + 32 * i, msgs + 32 * i, sigs + 96 * i) != 0)
      __atomic_store_n(&fail, -1, __ATOMIC_RELAXED);
  }
  return fail;
}

int m3x_oracle_bls_h2c_g2(const uint8_t msg[32], uint8_t out_uncomp[192]) {
  bls_init();
  g2_aff_t h;
  h2c_g2(&h, msg);
  g2_to_uncomp(&h, out_uncomp);
  return 0;
}

int m3x_oracle_bls_pairing(const uint8_t p_uncomp[96],
                           const uint8_t q_uncomp[192], uint8_t out[576]) {
  bls_init();
  g1_aff_t p;
  g2_aff_t q;
  if (g1_from_uncomp(&p, p_uncomp) != 0) return -1;
  if (g2_from_uncomp(&q, q_uncomp) != 0) return -1;
  fp12_t f, e;
  fp12_one(&f);
  miller(&f, &p, &q);
  final_exp(&e, &f);
  fp12_to_bytes(&e, out);
  return 0;
}

int m3x_oracle_bls_g1_mul(const uint8_t p_uncomp[96],
                          const uint8_t scalar_be[32], uint8_t out[96]) {
  bls_init();
  g1_aff_t p;
  if (g1_from_uncomp(&p, p_uncomp) != 0) return -1;
  g1_jac_t j;
  g1_mul_be(&j, &p, scalar_be, 32);
  g1_aff_t a;
  g1_to_aff(&a, &j);
  g1_to_uncomp(&a, out);
  return 0;
}

int m3x_oracle_g2_subgroup_check(const uint8_t uncomp[192], int use_ref) {
  /* introspection for tests: order-r membership of an uncompressed
   * E'(Fp2) point, via the fast psi criterion (use_ref=0) or the
   * [r]Q reference form (use_ref=1). Returns 1 in-subgroup, 0 not,
   * <0 malformed/off-curve. */
  bls_init();
  g2_aff_t p;
  if (g2_from_uncomp(&p, uncomp) != 0) return -1;
  return use_ref ? g2_in_subgroup_ref(&p) : g2_in_subgroup(&p);
}

int m3x_oracle_bls_selftest(void) {
  /* cross-checks of the round-2 fast paths against the reference forms
   * (returns 0 ok, <0 which check failed):
   *  1. final_exp3(f) == final_exp(f)^3
   *  2. final_exp(miller_jacobian) == final_exp(miller_affine_ref)
   *  3. psi subgroup check agrees with [r]Q on valid points */
  bls_init();
  uint8_t msg[32];
  for (int i = 0; i < 32; i++) msg[i] = (uint8_t)(0x33 + i);
  g2_aff_t h;
  h2c_g2(&h, msg);
  fp12_t f, fr;
  fp12_one(&f);
  miller(&f, &G1_GEN, &h);
  fp12_one(&fr);
  miller_affine_ref(&fr, &G1_GEN, &h);
  fp12_t e1, e2, e3, t;
  final_exp(&e1, &f);
  final_exp(&e2, &fr);
  /* 2: jacobian vs affine miller agree after final exp */
  if (memcmp(&e1, &e2, sizeof(e1)) != 0) return -2;
  /* 1: cubed chain == standard cubed */
  final_exp3(&e3, &f);
  fp12_mul_(&t, &e1, &e1);
  fp12_mul_(&t, &t, &e1);
  if (memcmp(&e3, &t, sizeof(t)) != 0) return -1;
  /* 3: subgroup-check agreement on a valid point */
  if (g2_in_subgroup(&h) != 1 || g2_in_subgroup_ref(&h) != 1) return -3;
  if (g2_in_subgroup(&G2_GEN) != 1 || g2_in_subgroup_ref(&G2_GEN) != 1)
    return -3;
  return 0;
}
