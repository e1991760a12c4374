/* internal shared declarations between bls12_381.c and batch_verify.c
 * (oracle only — see oracle/oracle.h header for usage rules). */
#ifndef M3X_BLS_INTERNAL_H
#define M3X_BLS_INTERNAL_H
#include <stdint.h>

typedef struct { uint64_t v[6]; } fp_t;          /* Montgomery form */
typedef struct { fp_t c0, c1; } fp2_t;           /* c0 + c1*u, u^2=-1 */
typedef struct { fp2_t c[6]; } fp12_t;           /* Fp2[w]/(w^6 - (1+u)) */
typedef struct { fp_t x, y; int inf; } g1_aff_t;
typedef struct { fp_t x, y, z; } g1_jac_t;       /* z==0 -> infinity */
typedef struct { fp2_t x, y; int inf; } g2_aff_t;
typedef struct { fp2_t x, y, z; } g2_jac_t;

void bls_init(void);

void fp_mul_(fp_t *r, const fp_t *a, const fp_t *b);
void fp_add_(fp_t *r, const fp_t *a, const fp_t *b);
void fp_sub_(fp_t *r, const fp_t *a, const fp_t *b);
int fp_is_zero_(const fp_t *a);
void fp_from_be48(fp_t *r, const uint8_t b[48], int *ok);
void fp_to_be48(const fp_t *a, uint8_t b[48]);

void fp2_mul_(fp2_t *r, const fp2_t *a, const fp2_t *b);

extern g1_aff_t G1_GEN;
extern g2_aff_t G2_GEN;
extern uint8_t ORDER_BE[32];

/* point ops */
void g1_dbl(g1_jac_t *r, const g1_jac_t *p);
void g1_addj(g1_jac_t *r, const g1_jac_t *p, const g1_jac_t *q);
void g1_add_aff(g1_jac_t *r, const g1_jac_t *p, const g1_aff_t *q);
void g1_from_aff(g1_jac_t *r, const g1_aff_t *a);
void g1_to_aff(g1_aff_t *r, const g1_jac_t *p);
void g1_mul_be(g1_jac_t *r, const g1_aff_t *p, const uint8_t *scalar_be,
               int nbytes);
int g1_jac_is_inf(const g1_jac_t *p);
void g2_dbl(g2_jac_t *r, const g2_jac_t *p);
void g2_addj(g2_jac_t *r, const g2_jac_t *p, const g2_jac_t *q);
void g2_from_aff(g2_jac_t *r, const g2_aff_t *a);
void g2_to_aff(g2_aff_t *r, const g2_jac_t *p);
void g2_mul_be(g2_jac_t *r, const g2_aff_t *p, const uint8_t *scalar_be,
               int nbytes);
int g2_jac_is_inf(const g2_jac_t *p);
int g2_in_subgroup(const g2_aff_t *p);
int g1_in_subgroup(const g1_aff_t *p);
int g1_on_curve(const g1_aff_t *p);
int g2_on_curve(const g2_aff_t *p);

/* serialization */
int g1_decompress(g1_aff_t *r, const uint8_t in[48]);
void g1_compress(const g1_aff_t *p, uint8_t out[48]);
void g1_to_uncomp(const g1_aff_t *p, uint8_t out[96]);
int g1_from_uncomp(g1_aff_t *r, const uint8_t in[96]);
int g2_decompress(g2_aff_t *r, const uint8_t in[96]);
void g2_compress(const g2_aff_t *p, uint8_t out[96]);
void g2_to_uncomp(const g2_aff_t *p, uint8_t out[192]);
int g2_from_uncomp(g2_aff_t *r, const uint8_t in[192]);

/* pairing */
void fp12_one(fp12_t *r);
void fp12_mul_(fp12_t *r, const fp12_t *a, const fp12_t *b);
int fp12_is_one(const fp12_t *a);
void miller(fp12_t *f, const g1_aff_t *p, const g2_aff_t *q); /* f *= ML(p,q) */
/* slow reference forms (cross-check tests only) */
void miller_affine_ref(fp12_t *f, const g1_aff_t *p, const g2_aff_t *q);
int g2_in_subgroup_ref(const g2_aff_t *p);
void final_exp(fp12_t *r, const fp12_t *f);
void final_exp3(fp12_t *r, const fp12_t *f); /* cubed hard part; ==1-equivalent */
void fp12_to_bytes(const fp12_t *a, uint8_t out[576]);

/* hash-to-curve (RFC 9380, G2 suite, DST = blst.rs:15) */
void h2c_g2(g2_aff_t *r, const uint8_t msg[32]);

#endif
