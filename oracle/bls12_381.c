/* BLS12-381 arithmetic for the CPU oracle — see bls12_381.h header comment
 * for what this restates and how it is pinned. TEST INFRASTRUCTURE ONLY. */
#include "bls_internal.h"
#include "bls_consts.h"
#include "oracle.h"
#include <string.h>

/* ------------------------------------------------------------------ Fp --- */

static fp_t FP_ZERO_, FP_ONE_, FP_R2_;
static uint64_t P_HALF[6]; /* (p-1)/2, standard form */
static uint64_t EXP_PM2[6], EXP_SQRT[6]; /* p-2, (p+1)/4 */

static int ge_p(const uint64_t a[6]) {
  for (int i = 5; i >= 0; i--) {
    if (a[i] > BLS_P[i]) return 1;
    if (a[i] < BLS_P[i]) return 0;
  }
  return 1;
}

static void sub_p(uint64_t a[6]) {
  unsigned __int128 bw = 0;
  for (int i = 0; i < 6; i++) {
    unsigned __int128 t = (unsigned __int128)a[i] - BLS_P[i] - (uint64_t)bw;
    a[i] = (uint64_t)t;
    bw = (t >> 64) & 1; /* borrow */
  }
}

void fp_add_(fp_t *r, const fp_t *a, const fp_t *b) {
  unsigned __int128 c = 0;
  uint64_t t[6];
  for (int i = 0; i < 6; i++) {
    c += (unsigned __int128)a->v[i] + b->v[i];
    t[i] = (uint64_t)c;
    c >>= 64;
  }
  if (c || ge_p(t)) sub_p(t);
  memcpy(r->v, t, 48);
}

void fp_sub_(fp_t *r, const fp_t *a, const fp_t *b) {
  unsigned __int128 bw = 0;
  uint64_t t[6];
  for (int i = 0; i < 6; i++) {
    unsigned __int128 x = (unsigned __int128)a->v[i] - b->v[i] - (uint64_t)bw;
    t[i] = (uint64_t)x;
    bw = (x >> 64) & 1;
  }
  if (bw) { /* add p back */
    unsigned __int128 c = 0;
    for (int i = 0; i < 6; i++) {
      c += (unsigned __int128)t[i] + BLS_P[i];
      t[i] = (uint64_t)c;
      c >>= 64;
    }
  }
  memcpy(r->v, t, 48);
}

static void fp_neg(fp_t *r, const fp_t *a) { fp_sub_(r, &FP_ZERO_, a); }

void fp_mul_(fp_t *r, const fp_t *a, const fp_t *b) {
  /* CIOS Montgomery multiplication, 6x64 limbs — fully unrolled so the
   * compiler schedules the mulx/carry chains (the rolled loop form costs
   * ~2.5x; measured round 2). */
  uint64_t t0 = 0, t1 = 0, t2 = 0, t3 = 0, t4 = 0, t5 = 0, t6 = 0, t7 = 0;
  const uint64_t *A = a->v, *B = b->v;
  uint64_t c, m;
#define M3X_MAC(hi, lo, x, y, add0, add1)                                      \
  do {                                                                         \
    unsigned __int128 _p = (unsigned __int128)(x) * (y) + (add0) + (add1);     \
    (lo) = (uint64_t)_p;                                                       \
    (hi) = (uint64_t)(_p >> 64);                                               \
  } while (0)
#define M3X_ROUND(bi)                                                          \
  do {                                                                         \
    M3X_MAC(c, t0, A[0], bi, t0, 0);                                           \
    M3X_MAC(c, t1, A[1], bi, t1, c);                                           \
    M3X_MAC(c, t2, A[2], bi, t2, c);                                           \
    M3X_MAC(c, t3, A[3], bi, t3, c);                                           \
    M3X_MAC(c, t4, A[4], bi, t4, c);                                           \
    M3X_MAC(c, t5, A[5], bi, t5, c);                                           \
    {                                                                          \
      unsigned __int128 s = (unsigned __int128)t6 + c;                         \
      t6 = (uint64_t)s;                                                        \
      t7 = (uint64_t)(s >> 64);                                                \
    }                                                                          \
    m = t0 * BLS_N0;                                                           \
    {                                                                          \
      unsigned __int128 p0 = (unsigned __int128)m * BLS_P[0] + t0;             \
      c = (uint64_t)(p0 >> 64);                                                \
    }                                                                          \
    M3X_MAC(c, t0, m, BLS_P[1], t1, c);                                        \
    M3X_MAC(c, t1, m, BLS_P[2], t2, c);                                        \
    M3X_MAC(c, t2, m, BLS_P[3], t3, c);                                        \
    M3X_MAC(c, t3, m, BLS_P[4], t4, c);                                        \
    M3X_MAC(c, t4, m, BLS_P[5], t5, c);                                        \
    {                                                                          \
      unsigned __int128 s = (unsigned __int128)t6 + c;                         \
      t5 = (uint64_t)s;                                                        \
      t6 = t7 + (uint64_t)(s >> 64);                                           \
    }                                                                          \
  } while (0)
  M3X_ROUND(B[0]);
  M3X_ROUND(B[1]);
  M3X_ROUND(B[2]);
  M3X_ROUND(B[3]);
  M3X_ROUND(B[4]);
  M3X_ROUND(B[5]);
#undef M3X_ROUND
#undef M3X_MAC
  uint64_t t[6] = {t0, t1, t2, t3, t4, t5};
  if (t6 || ge_p(t)) sub_p(t);
  memcpy(r->v, t, 48);
}

/* specialized squaring, fully unrolled SOS: 21 products + doubling vs the
 * multiply's 36 (13% faster measured; bit-exact over 300k chained) — the
 * sqrt/inv pow chains are squaring-dominated. */
#define M3X_SQMAC(hi, lo, x, y, a0, a1)                                        \
  do {                                                                         \
    unsigned __int128 _p = (unsigned __int128)(x) * (y) + (a0) + (a1);         \
    (lo) = (uint64_t)_p;                                                       \
    (hi) = (uint64_t)(_p >> 64);                                               \
  } while (0)

static void fp_sqr(fp_t *r, const fp_t *a) {
  const uint64_t *A = a->v;
  uint64_t t1,t2,t3,t4,t5,t6,t7,t8,t9,t10,c;
  /* row i=0: j=1..5 */
  M3X_SQMAC(c, t1, A[0], A[1], 0, 0);
  M3X_SQMAC(c, t2, A[0], A[2], 0, c);
  M3X_SQMAC(c, t3, A[0], A[3], 0, c);
  M3X_SQMAC(c, t4, A[0], A[4], 0, c);
  M3X_SQMAC(c, t5, A[0], A[5], 0, c);
  t6 = c;
  /* row i=1: j=2..5 */
  M3X_SQMAC(c, t3, A[1], A[2], t3, 0);
  M3X_SQMAC(c, t4, A[1], A[3], t4, c);
  M3X_SQMAC(c, t5, A[1], A[4], t5, c);
  M3X_SQMAC(c, t6, A[1], A[5], t6, c);
  t7 = c;
  /* row i=2 */
  M3X_SQMAC(c, t5, A[2], A[3], t5, 0);
  M3X_SQMAC(c, t6, A[2], A[4], t6, c);
  M3X_SQMAC(c, t7, A[2], A[5], t7, c);
  t8 = c;
  /* row i=3 */
  M3X_SQMAC(c, t7, A[3], A[4], t7, 0);
  M3X_SQMAC(c, t8, A[3], A[5], t8, c);
  t9 = c;
  /* row i=4 */
  M3X_SQMAC(c, t9, A[4], A[5], t9, 0);
  t10 = c;
  /* double + diagonals */
  uint64_t w[13];
  w[0]=0; w[1]=t1; w[2]=t2; w[3]=t3; w[4]=t4; w[5]=t5; w[6]=t6; w[7]=t7; w[8]=t8; w[9]=t9; w[10]=t10; w[11]=0; w[12]=0;
  uint64_t cc = 0;
  for (int i = 1; i <= 11; i++) { uint64_t nv = (w[i] << 1) | cc; cc = w[i] >> 63; w[i] = nv; }
  unsigned __int128 p; uint64_t carry = 0;
  for (int i = 0; i < 6; i++) {
    p = (unsigned __int128)A[i]*A[i] + w[2*i] + carry;
    w[2*i] = (uint64_t)p;
    unsigned __int128 p2 = (unsigned __int128)w[2*i+1] + (uint64_t)(p >> 64);
    w[2*i+1] = (uint64_t)p2;
    carry = (uint64_t)(p2 >> 64);
  }
  /* Montgomery reduce */
  uint64_t s0=w[0],s1=w[1],s2=w[2],s3=w[3],s4=w[4],s5=w[5];
  uint64_t hc = 0; uint64_t m;
#define REDR(hiw) do { \
    m = s0 * BLS_N0; \
    { unsigned __int128 p0 = (unsigned __int128)m * BLS_P[0] + s0; c = (uint64_t)(p0 >> 64); } \
    M3X_SQMAC(c, s0, m, BLS_P[1], s1, c); \
    M3X_SQMAC(c, s1, m, BLS_P[2], s2, c); \
    M3X_SQMAC(c, s2, m, BLS_P[3], s3, c); \
    M3X_SQMAC(c, s3, m, BLS_P[4], s4, c); \
    M3X_SQMAC(c, s4, m, BLS_P[5], s5, c); \
    { unsigned __int128 s = (unsigned __int128)(hiw) + c + hc; s5 = (uint64_t)s; hc = (uint64_t)(s >> 64); } \
  } while (0)
  REDR(w[6]); REDR(w[7]); REDR(w[8]); REDR(w[9]); REDR(w[10]); REDR(w[11]);
#undef REDR
  uint64_t out[6] = {s0,s1,s2,s3,s4,s5};
  if (hc || ge_p(out)) sub_p(out);
  memcpy(r->v, out, 48);
}
#undef M3X_SQMAC


/* ---- lazy-reduction support (round 2): 12-limb full product + one
 * Montgomery reduction, so fp2_mul does 3 wide muls + 2 reductions
 * instead of 3 full CIOS (≈17% fewer MACs; bounds checked below). ---- */

typedef struct { uint64_t w[12]; } fp_wide_t;

/* t = a*b, full 768-bit product (no reduction) */
static void fp_mul_wide(fp_wide_t *t, const fp_t *a, const fp_t *b) {
  uint64_t w0=0,w1=0,w2=0,w3=0,w4=0,w5=0,w6=0,w7=0,w8=0,w9=0,w10=0,w11=0;
  const uint64_t *A = a->v, *B = b->v;
  uint64_t c;
#define M3X_MACW(hi, lo, x, y, add0, add1)                                     \
  do {                                                                         \
    unsigned __int128 _p = (unsigned __int128)(x) * (y) + (add0) + (add1);     \
    (lo) = (uint64_t)_p;                                                       \
    (hi) = (uint64_t)(_p >> 64);                                               \
  } while (0)
#define M3X_ROWW(i, r0, r1, r2, r3, r4, r5, r6)                                \
  do {                                                                         \
    M3X_MACW(c, r0, A[0], B[i], r0, 0);                                        \
    M3X_MACW(c, r1, A[1], B[i], r1, c);                                        \
    M3X_MACW(c, r2, A[2], B[i], r2, c);                                        \
    M3X_MACW(c, r3, A[3], B[i], r3, c);                                        \
    M3X_MACW(c, r4, A[4], B[i], r4, c);                                        \
    M3X_MACW(c, r5, A[5], B[i], r5, c);                                        \
    r6 = c;                                                                    \
  } while (0)
  M3X_ROWW(0, w0, w1, w2, w3, w4, w5, w6);
  M3X_ROWW(1, w1, w2, w3, w4, w5, w6, w7);
  M3X_ROWW(2, w2, w3, w4, w5, w6, w7, w8);
  M3X_ROWW(3, w3, w4, w5, w6, w7, w8, w9);
  M3X_ROWW(4, w4, w5, w6, w7, w8, w9, w10);
  M3X_ROWW(5, w5, w6, w7, w8, w9, w10, w11);
#undef M3X_ROWW
  t->w[0]=w0; t->w[1]=w1; t->w[2]=w2; t->w[3]=w3; t->w[4]=w4; t->w[5]=w5;
  t->w[6]=w6; t->w[7]=w7; t->w[8]=w8; t->w[9]=w9; t->w[10]=w10; t->w[11]=w11;
}

/* wide add/sub (no reduction); caller tracks bounds */
static void fp_wide_add(fp_wide_t *r, const fp_wide_t *a, const fp_wide_t *b) {
  unsigned __int128 cc = 0;
  for (int i = 0; i < 12; i++) {
    cc += (unsigned __int128)a->w[i] + b->w[i];
    r->w[i] = (uint64_t)cc;
    cc >>= 64;
  }
}

/* r = a - b + 2p^2 (keeps the value positive; 2p^2 < 2^763) */
static uint64_t PP2_[12]; /* 2*p^2, set in bls_init */
static void fp_wide_sub_pp2(fp_wide_t *r, const fp_wide_t *a,
                            const fp_wide_t *b) {
  unsigned __int128 cc = 0;
  uint64_t t[12];
  for (int i = 0; i < 12; i++) {
    cc += (unsigned __int128)a->w[i] + PP2_[i];
    t[i] = (uint64_t)cc;
    cc >>= 64;
  }
  unsigned __int128 bw = 0;
  for (int i = 0; i < 12; i++) {
    unsigned __int128 x = (unsigned __int128)t[i] - b->w[i] - (uint64_t)bw;
    r->w[i] = (uint64_t)x;
    bw = (x >> 64) & 1;
  }
}

/* Montgomery reduction of a 12-limb T < p*2^384: r = T*R^-1 mod p */
static void fp_redc(fp_t *r, const fp_wide_t *T) {
  uint64_t t0=T->w[0],t1=T->w[1],t2=T->w[2],t3=T->w[3],t4=T->w[4],t5=T->w[5];
  uint64_t hi6=T->w[6],hi7=T->w[7],hi8=T->w[8],hi9=T->w[9],hi10=T->w[10],hi11=T->w[11];
  uint64_t carry6 = 0; /* accumulated carries into the high half */
  uint64_t c, m;
#define M3X_REDR()                                                             \
  do {                                                                         \
    m = t0 * BLS_N0;                                                           \
    {                                                                          \
      unsigned __int128 p0 = (unsigned __int128)m * BLS_P[0] + t0;             \
      c = (uint64_t)(p0 >> 64);                                                \
    }                                                                          \
    M3X_MACW(c, t0, m, BLS_P[1], t1, c);                                       \
    M3X_MACW(c, t1, m, BLS_P[2], t2, c);                                       \
    M3X_MACW(c, t2, m, BLS_P[3], t3, c);                                       \
    M3X_MACW(c, t3, m, BLS_P[4], t4, c);                                       \
    M3X_MACW(c, t4, m, BLS_P[5], t5, c);                                       \
  } while (0)
  /* 6 rounds; after each, shift in the next high limb + carry */
  unsigned __int128 s;
  M3X_REDR(); s = (unsigned __int128)hi6 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
  M3X_REDR(); s = (unsigned __int128)hi7 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
  M3X_REDR(); s = (unsigned __int128)hi8 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
  M3X_REDR(); s = (unsigned __int128)hi9 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
  M3X_REDR(); s = (unsigned __int128)hi10 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
  M3X_REDR(); s = (unsigned __int128)hi11 + c + carry6; t5 = (uint64_t)s; carry6 = (uint64_t)(s >> 64);
#undef M3X_REDR
  uint64_t t[6] = {t0, t1, t2, t3, t4, t5};
  if (carry6 || ge_p(t)) sub_p(t);
  memcpy(r->v, t, 48);
}

int fp_is_zero_(const fp_t *a) {
  uint64_t o = 0;
  for (int i = 0; i < 6; i++) o |= a->v[i];
  return o == 0;
}

static int fp_eq(const fp_t *a, const fp_t *b) {
  return memcmp(a->v, b->v, 48) == 0;
}

static void fp_from_std(fp_t *r, const uint64_t std[6]) {
  fp_t s;
  memcpy(s.v, std, 48);
  memcpy(r->v, FP_R2_.v, 48);
  fp_mul_(r, &s, &FP_R2_); /* to Montgomery */
}

static void fp_to_std(uint64_t std[6], const fp_t *a) {
  fp_t one_std = {{1, 0, 0, 0, 0, 0}};
  fp_t t;
  fp_mul_(&t, a, &one_std); /* from Montgomery */
  memcpy(std, t.v, 48);
}

/* MSB-first exponentiation by a little-endian limb exponent */
static void fp_pow_limbs(fp_t *r, const fp_t *a, const uint64_t *e, int n) {
  fp_t acc;
  memcpy(&acc, &FP_ONE_, sizeof(fp_t));
  int started = 0;
  for (int i = n - 1; i >= 0; i--) {
    for (int b = 63; b >= 0; b--) {
      if (started) fp_sqr(&acc, &acc);
      if ((e[i] >> b) & 1) {
        if (started)
          fp_mul_(&acc, &acc, a);
        else {
          memcpy(&acc, a, sizeof(fp_t));
          started = 1;
        }
      }
    }
  }
  memcpy(r, &acc, sizeof(fp_t));
}

static void fp_inv(fp_t *r, const fp_t *a) { fp_pow_limbs(r, a, EXP_PM2, 6); }

/* sqrt candidate a^((p+1)/4); caller must verify square */
static int fp_sqrt(fp_t *r, const fp_t *a) {
  fp_t s, s2;
  fp_pow_limbs(&s, a, EXP_SQRT, 6);
  fp_sqr(&s2, &s);
  if (!fp_eq(&s2, a)) return 0;
  memcpy(r, &s, sizeof(fp_t));
  return 1;
}

/* standard-form comparison with (p-1)/2: 1 if a_std > half */
static int fp_gt_half(const fp_t *a) {
  uint64_t s[6];
  fp_to_std(s, a);
  for (int i = 5; i >= 0; i--) {
    if (s[i] > P_HALF[i]) return 1;
    if (s[i] < P_HALF[i]) return 0;
  }
  return 0;
}

static int fp_is_odd_std(const fp_t *a) {
  uint64_t s[6];
  fp_to_std(s, a);
  return (int)(s[0] & 1);
}

void fp_to_be48(const fp_t *a, uint8_t b[48]) {
  uint64_t s[6];
  fp_to_std(s, a);
  for (int i = 0; i < 6; i++)
    for (int j = 0; j < 8; j++) b[8 * i + j] = (uint8_t)(s[5 - i] >> (56 - 8 * j));
}

void fp_from_be48(fp_t *r, const uint8_t b[48], int *ok) {
  uint64_t s[6] = {0};
  for (int i = 0; i < 6; i++)
    for (int j = 0; j < 8; j++)
      s[5 - i] = (s[5 - i] << 8) | b[8 * i + j];
  if (ge_p(s)) {
    *ok = 0;
    return;
  }
  fp_from_std(r, s);
  *ok = 1;
}

/* ----------------------------------------------------------------- Fp2 --- */

static fp2_t F2_ZERO_, F2_ONE_, XI_, XI_INV_, TWO_INV_;
static fp2_t SSWU_A_, SSWU_B_, SSWU_Z_, PSI_CX_, PSI_CY_;
static fp2_t FW1_POW[6], FW2_POW[6], Z6_POW[6];
static fp2_t ISO_KX[4], ISO_KXD[3], ISO_KY[4], ISO_KYD[4];

static void fp2_add(fp2_t *r, const fp2_t *a, const fp2_t *b) {
  fp_add_(&r->c0, &a->c0, &b->c0);
  fp_add_(&r->c1, &a->c1, &b->c1);
}
static void fp2_sub(fp2_t *r, const fp2_t *a, const fp2_t *b) {
  fp_sub_(&r->c0, &a->c0, &b->c0);
  fp_sub_(&r->c1, &a->c1, &b->c1);
}
static void fp2_neg(fp2_t *r, const fp2_t *a) {
  fp_neg(&r->c0, &a->c0);
  fp_neg(&r->c1, &a->c1);
}
void fp2_mul_(fp2_t *r, const fp2_t *a, const fp2_t *b) {
  /* Karatsuba with LAZY reduction: 3 wide products + 2 Montgomery
   * reductions (vs 3 full CIOS). Offsets by 2p^2 keep the wide
   * differences positive; all inputs < p so every reduction input
   * < 3p^2 < p*2^384 (single conditional subtract). */
  fp_wide_t t0, t1, m, s01, cw;
  fp_t sa, sb;
  fp_mul_wide(&t0, &a->c0, &b->c0);
  fp_mul_wide(&t1, &a->c1, &b->c1);
  fp_add_(&sa, &a->c0, &a->c1);
  fp_add_(&sb, &b->c0, &b->c1);
  fp_mul_wide(&m, &sa, &sb);
  fp_wide_sub_pp2(&cw, &t0, &t1); /* c0 = t0 - t1 */
  fp_redc(&r->c0, &cw);
  fp_wide_add(&s01, &t0, &t1);
  fp_wide_sub_pp2(&cw, &m, &s01); /* c1 = m - t0 - t1 */
  fp_redc(&r->c1, &cw);
}
static void fp2_sqr(fp2_t *r, const fp2_t *a) {
  fp_t s, d, m;
  fp_add_(&s, &a->c0, &a->c1);
  fp_sub_(&d, &a->c0, &a->c1);
  fp_mul_(&m, &a->c0, &a->c1);
  fp_mul_(&s, &s, &d);
  fp_add_(&r->c1, &m, &m);
  memcpy(&r->c0, &s, sizeof(fp_t));
}
static void fp2_conj(fp2_t *r, const fp2_t *a) {
  memcpy(&r->c0, &a->c0, sizeof(fp_t));
  fp_neg(&r->c1, &a->c1);
}
static int fp2_is_zero(const fp2_t *a) {
  return fp_is_zero_(&a->c0) && fp_is_zero_(&a->c1);
}
static int fp2_eq(const fp2_t *a, const fp2_t *b) {
  return fp_eq(&a->c0, &b->c0) && fp_eq(&a->c1, &b->c1);
}
static void fp2_inv(fp2_t *r, const fp2_t *a) {
  fp_t n, t0, t1;
  fp_sqr(&t0, &a->c0);
  fp_sqr(&t1, &a->c1);
  fp_add_(&n, &t0, &t1);
  fp_inv(&n, &n);
  fp_mul_(&r->c0, &a->c0, &n);
  fp_mul_(&t0, &a->c1, &n);
  fp_neg(&r->c1, &t0);
}
static void fp2_mul_fp(fp2_t *r, const fp2_t *a, const fp_t *k) {
  fp_mul_(&r->c0, &a->c0, k);
  fp_mul_(&r->c1, &a->c1, k);
}
static void fp2_dbl(fp2_t *r, const fp2_t *a) { fp2_add(r, a, a); }

/* sqrt in Fp2 (p = 3 mod 4), norm method; returns 0 if non-square */
static int fp2_sqrt(fp2_t *r, const fp2_t *a) {
  if (fp2_is_zero(a)) {
    memcpy(r, &F2_ZERO_, sizeof(fp2_t));
    return 1;
  }
  if (fp_is_zero_(&a->c1)) {
    fp_t s;
    if (fp_sqrt(&s, &a->c0)) {
      memcpy(&r->c0, &s, sizeof(fp_t));
      memcpy(&r->c1, &FP_ZERO_, sizeof(fp_t));
      return 1;
    }
    fp_t na;
    fp_neg(&na, &a->c0);
    if (!fp_sqrt(&s, &na)) return 0;
    memcpy(&r->c0, &FP_ZERO_, sizeof(fp_t));
    memcpy(&r->c1, &s, sizeof(fp_t));
    return 1;
  }
  fp_t n, s, d, x0, x1, t;
  fp_sqr(&n, &a->c0);
  fp_sqr(&t, &a->c1);
  fp_add_(&n, &n, &t);
  if (!fp_sqrt(&s, &n)) return 0;
  fp_add_(&d, &a->c0, &s);
  fp_mul_(&d, &d, &TWO_INV_.c0);
  if (!fp_sqrt(&x0, &d)) {
    fp_sub_(&d, &a->c0, &s);
    fp_mul_(&d, &d, &TWO_INV_.c0);
    if (!fp_sqrt(&x0, &d)) return 0;
  }
  fp_add_(&t, &x0, &x0);
  fp_inv(&t, &t);
  fp_mul_(&x1, &a->c1, &t);
  fp2_t cand = {{{0}}};
  memcpy(&cand.c0, &x0, sizeof(fp_t));
  memcpy(&cand.c1, &x1, sizeof(fp_t));
  fp2_t sq;
  fp2_sqr(&sq, &cand);
  if (!fp2_eq(&sq, a)) return 0;
  memcpy(r, &cand, sizeof(fp2_t));
  return 1;
}

/* lexicographic "y is largest": serialization compares c1 then c0
 * (generic_signature.rs encoding; ZCash flag rules) */
static int fp2_gt_half_lex(const fp2_t *y) {
  if (!fp_is_zero_(&y->c1)) return fp_gt_half(&y->c1);
  if (!fp_is_zero_(&y->c0)) return fp_gt_half(&y->c0);
  return 0;
}

static int fp2_sgn0(const fp2_t *x) {
  int s0 = fp_is_odd_std(&x->c0);
  int z0 = fp_is_zero_(&x->c0);
  int s1 = fp_is_odd_std(&x->c1);
  return s0 | (z0 & s1);
}

/* ----------------------------------------------------------- G1 points --- */

g1_aff_t G1_GEN;
g2_aff_t G2_GEN;
uint8_t ORDER_BE[32];
static fp_t B1_; /* 4 */
static fp2_t B2_; /* 4(1+u) */

int g1_jac_is_inf(const g1_jac_t *p) { return fp_is_zero_(&p->z); }

void g1_from_aff(g1_jac_t *r, const g1_aff_t *a) {
  if (a->inf) {
    memset(r, 0, sizeof(*r));
    return;
  }
  memcpy(&r->x, &a->x, sizeof(fp_t));
  memcpy(&r->y, &a->y, sizeof(fp_t));
  memcpy(&r->z, &FP_ONE_, sizeof(fp_t));
}

void g1_to_aff(g1_aff_t *r, const g1_jac_t *p) {
  if (g1_jac_is_inf(p)) {
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return;
  }
  fp_t zi, zi2, zi3;
  fp_inv(&zi, &p->z);
  fp_sqr(&zi2, &zi);
  fp_mul_(&zi3, &zi2, &zi);
  fp_mul_(&r->x, &p->x, &zi2);
  fp_mul_(&r->y, &p->y, &zi3);
  r->inf = 0;
}

void g1_dbl(g1_jac_t *r, const g1_jac_t *p) {
  if (g1_jac_is_inf(p)) {
    *r = *p;
    return;
  }
  fp_t A, B, C, D, E, F, t;
  fp_sqr(&A, &p->x);
  fp_sqr(&B, &p->y);
  fp_sqr(&C, &B);
  fp_add_(&D, &p->x, &B);
  fp_sqr(&D, &D);
  fp_sub_(&D, &D, &A);
  fp_sub_(&D, &D, &C);
  fp_add_(&D, &D, &D);
  fp_add_(&E, &A, &A);
  fp_add_(&E, &E, &A);
  fp_sqr(&F, &E);
  fp_sub_(&F, &F, &D);
  fp_sub_(&F, &F, &D); /* X3 */
  fp_mul_(&t, &p->y, &p->z);
  fp_add_(&r->z, &t, &t);
  fp_sub_(&t, &D, &F);
  fp_mul_(&t, &E, &t);
  fp_add_(&C, &C, &C);
  fp_add_(&C, &C, &C);
  fp_add_(&C, &C, &C); /* 8C */
  fp_sub_(&r->y, &t, &C);
  memcpy(&r->x, &F, sizeof(fp_t));
}

void g1_addj(g1_jac_t *r, const g1_jac_t *p, const g1_jac_t *q) {
  if (g1_jac_is_inf(p)) {
    *r = *q;
    return;
  }
  if (g1_jac_is_inf(q)) {
    *r = *p;
    return;
  }
  fp_t z1z1, z2z2, u1, u2, s1, s2, t;
  fp_sqr(&z1z1, &p->z);
  fp_sqr(&z2z2, &q->z);
  fp_mul_(&u1, &p->x, &z2z2);
  fp_mul_(&u2, &q->x, &z1z1);
  fp_mul_(&t, &q->z, &z2z2);
  fp_mul_(&s1, &p->y, &t);
  fp_mul_(&t, &p->z, &z1z1);
  fp_mul_(&s2, &q->y, &t);
  if (fp_eq(&u1, &u2)) {
    if (fp_eq(&s1, &s2)) {
      g1_dbl(r, p);
      return;
    }
    memset(r, 0, sizeof(*r));
    return;
  }
  fp_t h, i, j, rr, v;
  fp_sub_(&h, &u2, &u1);
  fp_add_(&i, &h, &h);
  fp_sqr(&i, &i);
  fp_mul_(&j, &h, &i);
  fp_sub_(&rr, &s2, &s1);
  fp_add_(&rr, &rr, &rr);
  fp_mul_(&v, &u1, &i);
  fp_sqr(&t, &rr);
  fp_sub_(&t, &t, &j);
  fp_sub_(&t, &t, &v);
  fp_sub_(&t, &t, &v); /* X3 */
  fp_t y3;
  fp_sub_(&y3, &v, &t);
  fp_mul_(&y3, &rr, &y3);
  fp_t s1j;
  fp_mul_(&s1j, &s1, &j);
  fp_add_(&s1j, &s1j, &s1j);
  fp_sub_(&y3, &y3, &s1j);
  fp_t z3;
  fp_add_(&z3, &p->z, &q->z);
  fp_sqr(&z3, &z3);
  fp_sub_(&z3, &z3, &z1z1);
  fp_sub_(&z3, &z3, &z2z2);
  fp_mul_(&z3, &z3, &h);
  memcpy(&r->x, &t, sizeof(fp_t));
  memcpy(&r->y, &y3, sizeof(fp_t));
  memcpy(&r->z, &z3, sizeof(fp_t));
}

void g1_add_aff(g1_jac_t *r, const g1_jac_t *p, const g1_aff_t *q) {
  /* mixed add (Jacobian += affine), madd-2007-bl: 7M+4S vs 11M+5S full */
  if (q->inf) {
    *r = *p;
    return;
  }
  if (g1_jac_is_inf(p)) {
    g1_from_aff(r, q);
    return;
  }
  fp_t z1z1, u2, s2, t;
  fp_sqr(&z1z1, &p->z);
  fp_mul_(&u2, &q->x, &z1z1);
  fp_mul_(&t, &p->z, &z1z1);
  fp_mul_(&s2, &q->y, &t);
  if (fp_eq(&u2, &p->x)) {
    if (fp_eq(&s2, &p->y)) {
      g1_dbl(r, p);
      return;
    }
    memset(r, 0, sizeof(*r));
    return;
  }
  fp_t h, hh, i, j, rr, v, x3, y3, z3;
  fp_sub_(&h, &u2, &p->x);
  fp_sqr(&hh, &h);
  fp_add_(&i, &hh, &hh);
  fp_add_(&i, &i, &i); /* 4*HH */
  fp_mul_(&j, &h, &i);
  fp_sub_(&rr, &s2, &p->y);
  fp_add_(&rr, &rr, &rr);
  fp_mul_(&v, &p->x, &i);
  fp_sqr(&x3, &rr);
  fp_sub_(&x3, &x3, &j);
  fp_sub_(&x3, &x3, &v);
  fp_sub_(&x3, &x3, &v);
  fp_sub_(&y3, &v, &x3);
  fp_mul_(&y3, &rr, &y3);
  fp_mul_(&t, &p->y, &j);
  fp_add_(&t, &t, &t);
  fp_sub_(&y3, &y3, &t);
  fp_add_(&z3, &p->z, &h);
  fp_sqr(&z3, &z3);
  fp_sub_(&z3, &z3, &z1z1);
  fp_sub_(&z3, &z3, &hh);
  r->x = x3;
  r->y = y3;
  r->z = z3;
}

void g1_mul_be(g1_jac_t *r, const g1_aff_t *p, const uint8_t *scalar_be,
               int nbytes) {
  g1_jac_t acc;
  memset(&acc, 0, sizeof(acc));
  for (int i = 0; i < nbytes; i++) {
    uint8_t byte = scalar_be[i];
    for (int b = 7; b >= 0; b--) {
      g1_dbl(&acc, &acc);
      if ((byte >> b) & 1) g1_add_aff(&acc, &acc, p);
    }
  }
  *r = acc;
}

int g1_on_curve(const g1_aff_t *p) {
  if (p->inf) return 1;
  fp_t l, rhs;
  fp_sqr(&l, &p->y);
  fp_sqr(&rhs, &p->x);
  fp_mul_(&rhs, &rhs, &p->x);
  fp_add_(&rhs, &rhs, &B1_);
  return fp_eq(&l, &rhs);
}

int g1_in_subgroup(const g1_aff_t *p) {
  if (p->inf) return 1;
  g1_jac_t t;
  g1_mul_be(&t, p, ORDER_BE, 32);
  return g1_jac_is_inf(&t);
}

/* ----------------------------------------------------------- G2 points --- */

int g2_jac_is_inf(const g2_jac_t *p) { return fp2_is_zero(&p->z); }

void g2_from_aff(g2_jac_t *r, const g2_aff_t *a) {
  if (a->inf) {
    memset(r, 0, sizeof(*r));
    return;
  }
  r->x = a->x;
  r->y = a->y;
  r->z = F2_ONE_;
}

void g2_to_aff(g2_aff_t *r, const g2_jac_t *p) {
  if (g2_jac_is_inf(p)) {
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return;
  }
  fp2_t zi, zi2, zi3;
  fp2_inv(&zi, &p->z);
  fp2_sqr(&zi2, &zi);
  fp2_mul_(&zi3, &zi2, &zi);
  fp2_mul_(&r->x, &p->x, &zi2);
  fp2_mul_(&r->y, &p->y, &zi3);
  r->inf = 0;
}

void g2_dbl(g2_jac_t *r, const g2_jac_t *p) {
  if (g2_jac_is_inf(p)) {
    *r = *p;
    return;
  }
  fp2_t A, B, C, D, E, F, t;
  fp2_sqr(&A, &p->x);
  fp2_sqr(&B, &p->y);
  fp2_sqr(&C, &B);
  fp2_add(&D, &p->x, &B);
  fp2_sqr(&D, &D);
  fp2_sub(&D, &D, &A);
  fp2_sub(&D, &D, &C);
  fp2_dbl(&D, &D);
  fp2_dbl(&E, &A);
  fp2_add(&E, &E, &A);
  fp2_sqr(&F, &E);
  fp2_sub(&F, &F, &D);
  fp2_sub(&F, &F, &D);
  fp2_mul_(&t, &p->y, &p->z);
  fp2_dbl(&r->z, &t);
  fp2_sub(&t, &D, &F);
  fp2_mul_(&t, &E, &t);
  fp2_dbl(&C, &C);
  fp2_dbl(&C, &C);
  fp2_dbl(&C, &C);
  fp2_sub(&r->y, &t, &C);
  r->x = F;
}

void g2_addj(g2_jac_t *r, const g2_jac_t *p, const g2_jac_t *q) {
  if (g2_jac_is_inf(p)) {
    *r = *q;
    return;
  }
  if (g2_jac_is_inf(q)) {
    *r = *p;
    return;
  }
  fp2_t z1z1, z2z2, u1, u2, s1, s2, t;
  fp2_sqr(&z1z1, &p->z);
  fp2_sqr(&z2z2, &q->z);
  fp2_mul_(&u1, &p->x, &z2z2);
  fp2_mul_(&u2, &q->x, &z1z1);
  fp2_mul_(&t, &q->z, &z2z2);
  fp2_mul_(&s1, &p->y, &t);
  fp2_mul_(&t, &p->z, &z1z1);
  fp2_mul_(&s2, &q->y, &t);
  if (fp2_eq(&u1, &u2)) {
    if (fp2_eq(&s1, &s2)) {
      g2_dbl(r, p);
      return;
    }
    memset(r, 0, sizeof(*r));
    return;
  }
  fp2_t h, i, j, rr, v;
  fp2_sub(&h, &u2, &u1);
  fp2_dbl(&i, &h);
  fp2_sqr(&i, &i);
  fp2_mul_(&j, &h, &i);
  fp2_sub(&rr, &s2, &s1);
  fp2_dbl(&rr, &rr);
  fp2_mul_(&v, &u1, &i);
  fp2_sqr(&t, &rr);
  fp2_sub(&t, &t, &j);
  fp2_sub(&t, &t, &v);
  fp2_sub(&t, &t, &v);
  fp2_t y3;
  fp2_sub(&y3, &v, &t);
  fp2_mul_(&y3, &rr, &y3);
  fp2_t s1j;
  fp2_mul_(&s1j, &s1, &j);
  fp2_dbl(&s1j, &s1j);
  fp2_sub(&y3, &y3, &s1j);
  fp2_t z3;
  fp2_add(&z3, &p->z, &q->z);
  fp2_sqr(&z3, &z3);
  fp2_sub(&z3, &z3, &z1z1);
  fp2_sub(&z3, &z3, &z2z2);
  fp2_mul_(&z3, &z3, &h);
  r->x = t;
  r->y = y3;
  r->z = z3;
}

void g2_mul_be(g2_jac_t *r, const g2_aff_t *p, const uint8_t *scalar_be,
               int nbytes) {
  g2_jac_t acc;
  memset(&acc, 0, sizeof(acc));
  g2_jac_t base;
  g2_from_aff(&base, p);
  for (int i = 0; i < nbytes; i++) {
    uint8_t byte = scalar_be[i];
    for (int b = 7; b >= 0; b--) {
      g2_dbl(&acc, &acc);
      if ((byte >> b) & 1) g2_addj(&acc, &acc, &base);
    }
  }
  *r = acc;
}

int g2_on_curve(const g2_aff_t *p) {
  if (p->inf) return 1;
  fp2_t l, rhs;
  fp2_sqr(&l, &p->y);
  fp2_sqr(&rhs, &p->x);
  fp2_mul_(&rhs, &rhs, &p->x);
  fp2_add(&rhs, &rhs, &B2_);
  return fp2_eq(&l, &rhs);
}

static void psi_g2(g2_aff_t *r, const g2_aff_t *p);
static void g2_mul_u64(g2_jac_t *r, const g2_aff_t *p, uint64_t k);

int g2_in_subgroup(const g2_aff_t *p) {
  /* fast check: psi(Q) == -[|x|]Q (x < 0) — same criterion as the GPU
   * kernel (bls_device.hh g2_in_subgroup; validated vs [r]Q in the
   * generator and cross-checked against g2_in_subgroup_ref in tests).
   * Compare affine psi(Q) with Jacobian -[|x|]Q by cross-multiplication. */
  if (p->inf) return 1;
  g2_aff_t ps;
  psi_g2(&ps, p);
  g2_jac_t xq;
  g2_mul_u64(&xq, p, BLS_X_ABS);
  if (g2_jac_is_inf(&xq)) return 0;
  fp2_t z2, z3, lx, ly, ny;
  fp2_sqr(&z2, &xq.z);
  fp2_mul_(&z3, &z2, &xq.z);
  fp2_mul_(&lx, &ps.x, &z2);
  fp2_neg(&ny, &xq.y);
  fp2_mul_(&ly, &ps.y, &z3);
  return fp2_eq(&lx, &xq.x) && fp2_eq(&ly, &ny);
}

int g2_in_subgroup_ref(const g2_aff_t *p) {
  if (p->inf) return 1;
  g2_jac_t t;
  g2_mul_be(&t, p, ORDER_BE, 32);
  return g2_jac_is_inf(&t);
}

/* -------------------------------------------------------- serialization --- */

int g1_decompress(g1_aff_t *r, const uint8_t in[48]) {
  uint8_t flags = in[0];
  if (!(flags & 0x80)) return -1;
  if (flags & 0x40) {
    if (flags & 0x20 || (flags & 0x1F)) return -1;
    for (int i = 1; i < 48; i++)
      if (in[i]) return -1;
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return 0;
  }
  uint8_t xb[48];
  memcpy(xb, in, 48);
  xb[0] &= 0x1F;
  int ok;
  fp_from_be48(&r->x, xb, &ok);
  if (!ok) return -1;
  fp_t rhs;
  fp_sqr(&rhs, &r->x);
  fp_mul_(&rhs, &rhs, &r->x);
  fp_add_(&rhs, &rhs, &B1_);
  if (!fp_sqrt(&r->y, &rhs)) return -1;
  if (fp_gt_half(&r->y) != !!(flags & 0x20)) fp_neg(&r->y, &r->y);
  r->inf = 0;
  return 0;
}

void g1_compress(const g1_aff_t *p, uint8_t out[48]) {
  if (p->inf) {
    memset(out, 0, 48);
    out[0] = 0xC0;
    return;
  }
  fp_to_be48(&p->x, out);
  out[0] |= 0x80;
  if (fp_gt_half(&p->y)) out[0] |= 0x20;
}

void g1_to_uncomp(const g1_aff_t *p, uint8_t out[96]) {
  if (p->inf) {
    memset(out, 0, 96);
    out[0] = 0x40;
    return;
  }
  fp_to_be48(&p->x, out);
  fp_to_be48(&p->y, out + 48);
}

int g1_from_uncomp(g1_aff_t *r, const uint8_t in[96]) {
  if (in[0] & 0x40) {
    for (int i = 0; i < 96; i++)
      if (in[i] != (i == 0 ? 0x40 : 0)) return -1;
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return 0;
  }
  int ok;
  fp_from_be48(&r->x, in, &ok);
  if (!ok) return -1;
  fp_from_be48(&r->y, in + 48, &ok);
  if (!ok) return -1;
  r->inf = 0;
  return g1_on_curve(r) ? 0 : -1;
}

int g2_decompress(g2_aff_t *r, const uint8_t in[96]) {
  uint8_t flags = in[0];
  if (!(flags & 0x80)) return -1;
  if (flags & 0x40) {
    if (flags & 0x20 || (flags & 0x1F)) return -1;
    for (int i = 1; i < 96; i++)
      if (in[i]) return -1;
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return 0;
  }
  uint8_t b[48];
  memcpy(b, in, 48);
  b[0] &= 0x1F;
  int ok;
  fp_from_be48(&r->x.c1, b, &ok); /* serialization is c1 || c0 */
  if (!ok) return -1;
  fp_from_be48(&r->x.c0, in + 48, &ok);
  if (!ok) return -1;
  fp2_t rhs;
  fp2_sqr(&rhs, &r->x);
  fp2_mul_(&rhs, &rhs, &r->x);
  fp2_add(&rhs, &rhs, &B2_);
  if (!fp2_sqrt(&r->y, &rhs)) return -1;
  if (fp2_gt_half_lex(&r->y) != !!(flags & 0x20)) fp2_neg(&r->y, &r->y);
  r->inf = 0;
  return 0;
}

void g2_compress(const g2_aff_t *p, uint8_t out[96]) {
  if (p->inf) {
    memset(out, 0, 96);
    out[0] = 0xC0;
    return;
  }
  fp_to_be48(&p->x.c1, out);
  fp_to_be48(&p->x.c0, out + 48);
  out[0] |= 0x80;
  if (fp2_gt_half_lex(&p->y)) out[0] |= 0x20;
}

void g2_to_uncomp(const g2_aff_t *p, uint8_t out[192]) {
  if (p->inf) {
    memset(out, 0, 192);
    out[0] = 0x40;
    return;
  }
  fp_to_be48(&p->x.c1, out);
  fp_to_be48(&p->x.c0, out + 48);
  fp_to_be48(&p->y.c1, out + 96);
  fp_to_be48(&p->y.c0, out + 144);
}

int g2_from_uncomp(g2_aff_t *r, const uint8_t in[192]) {
  if (in[0] & 0x40) {
    for (int i = 0; i < 192; i++)
      if (in[i] != (i == 0 ? 0x40 : 0)) return -1;
    memset(r, 0, sizeof(*r));
    r->inf = 1;
    return 0;
  }
  int ok;
  fp_from_be48(&r->x.c1, in, &ok);
  if (!ok) return -1;
  fp_from_be48(&r->x.c0, in + 48, &ok);
  if (!ok) return -1;
  fp_from_be48(&r->y.c1, in + 96, &ok);
  if (!ok) return -1;
  fp_from_be48(&r->y.c0, in + 144, &ok);
  if (!ok) return -1;
  r->inf = 0;
  return g2_on_curve(r) ? 0 : -1;
}

/* ---------------------------------------------------------------- Fp12 --- */

void fp12_one(fp12_t *r) {
  memset(r, 0, sizeof(*r));
  r->c[0] = F2_ONE_;
}

/* generic degree-11 convolution — kept as the reference form (fixture
 * tests assert fast == generic; also used for sparse operands) */
void fp12_mul_generic(fp12_t *r, const fp12_t *a, const fp12_t *b) {
  fp2_t acc[11];
  memset(acc, 0, sizeof(acc));
  for (int i = 0; i < 6; i++) {
    if (fp2_is_zero(&a->c[i])) continue;
    for (int j = 0; j < 6; j++) {
      if (fp2_is_zero(&b->c[j])) continue;
      fp2_t t;
      fp2_mul_(&t, &a->c[i], &b->c[j]);
      fp2_add(&acc[i + j], &acc[i + j], &t);
    }
  }
  for (int k = 10; k >= 6; k--) {
    fp2_t t;
    fp2_mul_(&t, &acc[k], &XI_);
    fp2_add(&acc[k - 6], &acc[k - 6], &t);
  }
  memcpy(r->c, acc, 6 * sizeof(fp2_t));
}

/* quadratic-over-cubic tower view: f = A + w*B with A = (c0,c2,c4),
 * B = (c1,c3,c5) in Fp6 = Fp2[v]/(v^3 - xi), w^2 = v. Same mapping as the
 * GPU header (index mapping validated against the schoolbook degree-6
 * multiply; fast==generic asserted in tests). */
typedef struct { fp2_t c[3]; } fp6_t;

static void f6_add(fp6_t *r, const fp6_t *a, const fp6_t *b) {
  fp2_add(&r->c[0], &a->c[0], &b->c[0]);
  fp2_add(&r->c[1], &a->c[1], &b->c[1]);
  fp2_add(&r->c[2], &a->c[2], &b->c[2]);
}
static void f6_sub(fp6_t *r, const fp6_t *a, const fp6_t *b) {
  fp2_sub(&r->c[0], &a->c[0], &b->c[0]);
  fp2_sub(&r->c[1], &a->c[1], &b->c[1]);
  fp2_sub(&r->c[2], &a->c[2], &b->c[2]);
}
static void f6_mul_v(fp6_t *r, const fp6_t *a) {
  fp2_t t;
  fp2_mul_(&t, &a->c[2], &XI_);
  r->c[2] = a->c[1];
  r->c[1] = a->c[0];
  r->c[0] = t;
}
static void f6_mul(fp6_t *r, const fp6_t *a, const fp6_t *b) {
  fp2_t acc[5], t;
  memset(acc, 0, sizeof(acc));
  for (int i = 0; i < 3; i++)
    for (int j = 0; j < 3; j++) {
      fp2_mul_(&t, &a->c[i], &b->c[j]);
      fp2_add(&acc[i + j], &acc[i + j], &t);
    }
  fp2_mul_(&t, &acc[3], &XI_);
  fp2_add(&r->c[0], &acc[0], &t);
  fp2_mul_(&t, &acc[4], &XI_);
  fp2_add(&r->c[1], &acc[1], &t);
  r->c[2] = acc[2];
}
static void f12_split(const fp12_t *f, fp6_t *A, fp6_t *B) {
  A->c[0] = f->c[0];
  A->c[1] = f->c[2];
  A->c[2] = f->c[4];
  B->c[0] = f->c[1];
  B->c[1] = f->c[3];
  B->c[2] = f->c[5];
}
static void f12_join(fp12_t *f, const fp6_t *A, const fp6_t *B) {
  f->c[0] = A->c[0];
  f->c[2] = A->c[1];
  f->c[4] = A->c[2];
  f->c[1] = B->c[0];
  f->c[3] = B->c[1];
  f->c[5] = B->c[2];
}

void fp12_mul_(fp12_t *r, const fp12_t *a, const fp12_t *b) {
  /* Karatsuba over the tower: 3 Fp6 muls = 27 fp2 muls (vs 36 schoolbook) */
  fp6_t A1, B1, A2, B2, aa, bb, s1, s2, cross, even, t;
  f12_split(a, &A1, &B1);
  f12_split(b, &A2, &B2);
  f6_mul(&aa, &A1, &A2);
  f6_mul(&bb, &B1, &B2);
  f6_add(&s1, &A1, &B1);
  f6_add(&s2, &A2, &B2);
  f6_mul(&cross, &s1, &s2);
  f6_sub(&cross, &cross, &aa);
  f6_sub(&cross, &cross, &bb); /* A1B2 + A2B1 */
  f6_mul_v(&t, &bb);
  f6_add(&even, &aa, &t); /* A1A2 + v B1B2 */
  f12_join(r, &even, &cross);
}

static void fp12_sqr(fp12_t *r, const fp12_t *a) {
  /* complex squaring: 2 Fp6 muls = 18 fp2 muls */
  fp6_t A, B, m1, t, u, even, odd;
  f12_split(a, &A, &B);
  f6_mul(&m1, &A, &B);
  f6_add(&t, &A, &B);
  f6_mul_v(&u, &B);
  f6_add(&u, &A, &u);
  f6_mul(&t, &t, &u); /* (A+B)(A+vB) */
  f6_sub(&t, &t, &m1);
  f6_mul_v(&u, &m1);
  f6_sub(&even, &t, &u); /* A^2 + v B^2 */
  f6_add(&odd, &m1, &m1); /* 2AB */
  f12_join(r, &even, &odd);
}

static void fp12_conj6(fp12_t *r, const fp12_t *a) {
  /* f^(p^6): odd w-coefficients negate (XI^((p^6-1)/6) == -1, asserted by
   * the generator) */
  for (int i = 0; i < 6; i++) {
    if (i % 2 == 0)
      r->c[i] = a->c[i];
    else
      fp2_neg(&r->c[i], &a->c[i]);
  }
}

static void fp12_frob_p(fp12_t *r, const fp12_t *a) {
  for (int i = 0; i < 6; i++) {
    fp2_t t;
    fp2_conj(&t, &a->c[i]);
    fp2_mul_(&r->c[i], &t, &FW1_POW[i]);
  }
}

static void fp12_frob_p2(fp12_t *r, const fp12_t *a) {
  for (int i = 0; i < 6; i++) fp2_mul_(&r->c[i], &a->c[i], &FW2_POW[i]);
}

static void fp12_inv(fp12_t *r, const fp12_t *a) {
  /* product of sigma-conjugates (w -> zeta6^i w); norm lands in Fp2 */
  fp12_t g, t;
  fp12_one(&g);
  for (int i = 1; i < 6; i++) {
    for (int j = 0; j < 6; j++) {
      fp2_t zij = Z6_POW[(i * j) % 6];
      fp2_mul_(&t.c[j], &a->c[j], &zij);
    }
    fp12_mul_(&g, &g, &t);
  }
  fp12_t n;
  fp12_mul_(&n, a, &g);
  fp2_t ninv;
  fp2_inv(&ninv, &n.c[0]);
  for (int j = 0; j < 6; j++) fp2_mul_(&r->c[j], &g.c[j], &ninv);
}

int fp12_is_one(const fp12_t *a) {
  if (!fp2_eq(&a->c[0], &F2_ONE_)) return 0;
  for (int i = 1; i < 6; i++)
    if (!fp2_is_zero(&a->c[i])) return 0;
  return 1;
}

static void fp12_pow_limbs(fp12_t *r, const fp12_t *a, const uint64_t *e,
                           int n) {
  fp12_t acc;
  fp12_one(&acc);
  int started = 0;
  for (int i = n - 1; i >= 0; i--) {
    for (int b = 63; b >= 0; b--) {
      if (started) fp12_sqr(&acc, &acc);
      if ((e[i] >> b) & 1) {
        if (started)
          fp12_mul_(&acc, &acc, a);
        else {
          *r = *a;
          acc = *a;
          started = 1;
        }
      }
    }
  }
  *r = acc;
}

void fp12_to_bytes(const fp12_t *a, uint8_t out[576]) {
  for (int i = 0; i < 6; i++) {
    fp_to_be48(&a->c[i].c0, out + 96 * i);
    fp_to_be48(&a->c[i].c1, out + 96 * i + 48);
  }
}

/* -------------------------------------------------------------- pairing --- */

/* multiply f by the sparse line  l = a0*w^0 + a3*w^3 + a5*w^5
 * (a0 in Fp embedded as Fp2). Validated against the generic Miller loop in
 * gen_bls_fixtures.py (sparse==generic check run in-session; re-checked by
 * tests against GT fixtures). */
static void fp12_mul_line(fp12_t *f, const fp2_t *a0, const fp2_t *a3,
                          const fp2_t *a5) {
  fp12_t l;
  memset(&l, 0, sizeof(l));
  l.c[0] = *a0;
  l.c[3] = *a3;
  l.c[5] = *a5;
  /* generic form zero-skips: 18 fp2 muls for the 3-coefficient line */
  fp12_mul_generic(f, f, &l);
}

/* Inversion-free Miller loop: T in Jacobian coordinates on the twist,
 * affine inputs (Zp = Zq = 1). Line scalings by Fp2-subfield factors
 * (Tz powers) vanish under the final exponentiation's easy part
 * (c^(p^6-1) = 1 for c in Fp2), so final_exp(miller(...)) is unchanged —
 * asserted against miller_affine_ref + the committed GT fixtures in
 * tests/test_oracle_bls.py. Formulas mirror the validated GPU
 * miller_raw (bls_device.hh) specialized to Zp = Zq = 1. */
void miller(fp12_t *f, const g1_aff_t *p, const g2_aff_t *q) {
  if (p->inf || q->inf) return; /* e(O,.) = e(.,O) = 1 */
  g2_jac_t T, qj;
  g2_from_aff(&T, q);
  g2_from_aff(&qj, q);
  fp12_t acc;
  fp12_one(&acc);
  for (int i = 62; i >= 0; i--) {
    fp12_sqr(&acc, &acc);
    /* doubling line from Jacobian T:
     * a0 = 2*Ty*Tz^3*yp ; a3 = (3Tx^3 - 2Ty^2)*xi^-1 ;
     * a5 = -3Tx^2*Tz^2*xp*xi^-1 */
    {
      fp2_t X2, Y2, Z2, Z3, a0, a3, a5, t, t2;
      fp2_sqr(&X2, &T.x);
      fp2_sqr(&Y2, &T.y);
      fp2_sqr(&Z2, &T.z);
      fp2_mul_(&Z3, &Z2, &T.z);
      fp2_mul_(&t, &T.y, &Z3);
      fp2_dbl(&t, &t);
      fp2_mul_fp(&a0, &t, &p->y);
      fp2_mul_(&t, &X2, &T.x);
      fp2_add(&t2, &t, &t);
      fp2_add(&t, &t, &t2); /* 3Tx^3 */
      fp2_dbl(&t2, &Y2);
      fp2_sub(&t, &t, &t2);
      fp2_mul_(&a3, &t, &XI_INV_);
      fp2_mul_(&t, &X2, &Z2);
      fp2_add(&t2, &t, &t);
      fp2_add(&t, &t, &t2); /* 3Tx^2Tz^2 */
      fp2_mul_fp(&t, &t, &p->x);
      fp2_neg(&t, &t);
      fp2_mul_(&a5, &t, &XI_INV_);
      fp12_mul_line(&acc, &a0, &a3, &a5);
      g2_dbl(&T, &T);
    }
    if ((BLS_X_ABS >> i) & 1) {
      /* addition line through Jacobian T and affine Q:
       * Hs = Tx - Qx*Tz^2 ; Ms = Ty - Qy*Tz^3 ;
       * a0 = Tz^3*Hs*yp ; a3 = (Ms*Tx - Ty*Hs)*xi^-1 ;
       * a5 = -Ms*Tz^2*xp*xi^-1 */
      fp2_t Z2, Z3, Hs, Ms, a0, a3, a5, t, t2;
      fp2_sqr(&Z2, &T.z);
      fp2_mul_(&Z3, &Z2, &T.z);
      fp2_mul_(&t, &q->x, &Z2);
      fp2_sub(&Hs, &T.x, &t);
      fp2_mul_(&t, &q->y, &Z3);
      fp2_sub(&Ms, &T.y, &t);
      fp2_mul_(&t, &Z3, &Hs);
      fp2_mul_fp(&a0, &t, &p->y);
      fp2_mul_(&t, &Ms, &T.x);
      fp2_mul_(&t2, &T.y, &Hs);
      fp2_sub(&t, &t, &t2);
      fp2_mul_(&a3, &t, &XI_INV_);
      fp2_mul_(&t, &Ms, &Z2);
      fp2_mul_fp(&t, &t, &p->x);
      fp2_neg(&t, &t);
      fp2_mul_(&a5, &t, &XI_INV_);
      fp12_mul_line(&acc, &a0, &a3, &a5);
      g2_addj(&T, &T, &qj);
    }
  }
  fp12_t conj;
  fp12_conj6(&conj, &acc); /* x < 0 */
  fp12_mul_(f, f, &conj);
}

/* original affine-lambda loop (per-step fp2_inv) — kept as the slow
 * reference form; tests assert final_exp-agreement with the Jacobian loop */
void miller_affine_ref(fp12_t *f, const g1_aff_t *p, const g2_aff_t *q) {
  if (p->inf || q->inf) return; /* e(O,.) = e(.,O) = 1 */
  /* T on the twist in affine Fp2; line coeffs:
   * l = yp + (lam*xT - yT)*xi^-1 * w^3 - lam*xp*xi^-1 * w^5 */
  fp2_t xT = q->x, yT = q->y;
  fp12_t acc;
  fp12_one(&acc);
  uint64_t c = BLS_X_ABS;
  int top = 63;
  while (!((c >> top) & 1)) top--;
  for (int i = top - 1; i >= 0; i--) {
    fp12_sqr(&acc, &acc);
    /* doubling: lam = 3 xT^2 / (2 yT) */
    fp2_t lam, t, den;
    fp2_sqr(&t, &xT);
    fp2_add(&lam, &t, &t);
    fp2_add(&lam, &lam, &t);
    fp2_dbl(&den, &yT);
    fp2_inv(&den, &den);
    fp2_mul_(&lam, &lam, &den);
    fp2_t a0, a3, a5;
    memset(&a0, 0, sizeof(a0));
    a0.c0 = p->y;
    fp2_mul_(&a3, &lam, &xT);
    fp2_sub(&a3, &a3, &yT);
    fp2_mul_(&a3, &a3, &XI_INV_);
    fp2_mul_fp(&a5, &lam, &p->x);
    fp2_neg(&a5, &a5);
    fp2_mul_(&a5, &a5, &XI_INV_);
    fp12_mul_line(&acc, &a0, &a3, &a5);
    /* T = 2T */
    fp2_t x3, y3;
    fp2_sqr(&x3, &lam);
    fp2_sub(&x3, &x3, &xT);
    fp2_sub(&x3, &x3, &xT);
    fp2_sub(&y3, &xT, &x3);
    fp2_mul_(&y3, &lam, &y3);
    fp2_sub(&y3, &y3, &yT);
    xT = x3;
    yT = y3;
    if ((c >> i) & 1) {
      /* addition: lam = (yQ - yT)/(xQ - xT) */
      fp2_sub(&lam, &q->y, &yT);
      fp2_sub(&den, &q->x, &xT);
      fp2_inv(&den, &den);
      fp2_mul_(&lam, &lam, &den);
      memset(&a0, 0, sizeof(a0));
      a0.c0 = p->y;
      fp2_mul_(&a3, &lam, &xT);
      fp2_sub(&a3, &a3, &yT);
      fp2_mul_(&a3, &a3, &XI_INV_);
      fp2_mul_fp(&a5, &lam, &p->x);
      fp2_neg(&a5, &a5);
      fp2_mul_(&a5, &a5, &XI_INV_);
      fp12_mul_line(&acc, &a0, &a3, &a5);
      fp2_sqr(&x3, &lam);
      fp2_sub(&x3, &x3, &xT);
      fp2_sub(&x3, &x3, &q->x);
      fp2_sub(&y3, &xT, &x3);
      fp2_mul_(&y3, &lam, &y3);
      fp2_sub(&y3, &y3, &yT);
      xT = x3;
      yT = y3;
    }
  }
  fp12_t conj;
  fp12_conj6(&conj, &acc); /* x < 0 */
  fp12_mul_(f, f, &conj);
}

void final_exp(fp12_t *r, const fp12_t *f) {
  fp12_t t, fi, e;
  fp12_conj6(&t, f);
  fp12_inv(&fi, f);
  fp12_mul_(&e, &t, &fi); /* f^(p^6 - 1) */
  fp12_frob_p2(&t, &e);
  fp12_mul_(&e, &t, &e); /* ^(p^2 + 1) */
  fp12_pow_limbs(r, &e, FINAL_EXP_D, FINAL_EXP_D_LIMBS); /* hard part */
}

static void fp12_pow_xabs(fp12_t *r, const fp12_t *a) {
  fp12_t acc = *a; /* top bit of |x| (bit 63) folded into the start */
  for (int b = 62; b >= 0; b--) {
    fp12_sqr(&acc, &acc);
    if ((BLS_X_ABS >> b) & 1) fp12_mul_(&acc, &acc, a);
  }
  *r = acc;
}

/* final_exp cubed: hard part via the (x-1)^2(x+p)(x^2+p^2-1)+3 addition
 * chain (~10x cheaper than the generic 1264-bit pow). Computes
 * final_exp(f)^3 — equivalent for the ==1 verdict since gcd(3, r) = 1
 * (same chain as the GPU final_exp_w; final_exp3(f) == final_exp(f)^3 is
 * asserted in tests). conj6 is the cyclotomic inverse after the easy
 * part. */
void final_exp3(fp12_t *r, const fp12_t *f) {
  fp12_t t, fi, e;
  fp12_conj6(&t, f);
  fp12_inv(&fi, f);
  fp12_mul_(&e, &t, &fi); /* f^(p^6 - 1) */
  fp12_frob_p2(&t, &e);
  fp12_mul_(&e, &t, &e); /* cyclotomic e */
  fp12_t u, v, w1, w2, s;
  fp12_pow_xabs(&s, &e);
  fp12_mul_(&s, &s, &e);
  fp12_conj6(&u, &s); /* u = e^(x-1)   (x = -|x|) */
  fp12_pow_xabs(&s, &u);
  fp12_mul_(&s, &s, &u);
  fp12_conj6(&v, &s); /* v = e^((x-1)^2) */
  fp12_pow_xabs(&s, &v);
  fp12_conj6(&s, &s); /* v^x */
  fp12_frob_p(&t, &v);
  fp12_mul_(&w1, &s, &t); /* w1 = v^(x+p) */
  fp12_pow_xabs(&s, &w1);
  fp12_conj6(&s, &s); /* w1^x */
  fp12_pow_xabs(&t, &s);
  fp12_conj6(&t, &t); /* w1^(x^2) */
  fp12_frob_p2(&s, &w1);
  fp12_mul_(&t, &t, &s);
  fp12_conj6(&s, &w1); /* w1^-1 */
  fp12_mul_(&w2, &t, &s); /* w2 = w1^(x^2+p^2-1) */
  fp12_sqr(&s, &e);
  fp12_mul_(&s, &s, &e); /* e^3 */
  fp12_mul_(r, &w2, &s);
}

/* -------------------------------------------------------- hash-to-curve --- */

static void expand_message_xmd(const uint8_t *msg, uint32_t msg_len,
                               uint8_t out[256]) {
  /* DST = blst.rs:15; SHA-256; len_in_bytes = 256, ell = 8 */
  static const uint8_t DST[] = "BLS_SIG_BLS12381G2_XMD:SHA-256_SSWU_RO_POP_";
  const uint32_t dst_len = sizeof(DST) - 1;
  uint8_t buf[64 + 64 + 2 + 1 + sizeof(DST)];
  uint32_t off = 0;
  memset(buf, 0, 64);
  off = 64;
  memcpy(buf + off, msg, msg_len);
  off += msg_len;
  buf[off++] = 1; /* I2OSP(256,2) = 0x01 0x00 */
  buf[off++] = 0;
  buf[off++] = 0; /* block counter 0 */
  memcpy(buf + off, DST, dst_len);
  off += dst_len;
  buf[off++] = (uint8_t)dst_len;
  uint8_t b0[32];
  m3x_oracle_sha256(buf, off, b0);
  uint8_t cur[32 + 1 + sizeof(DST)];
  memcpy(cur, b0, 32);
  cur[32] = 1;
  memcpy(cur + 33, DST, dst_len);
  cur[33 + dst_len] = (uint8_t)dst_len;
  uint8_t bi[32];
  m3x_oracle_sha256(cur, 33 + dst_len + 1, bi);
  memcpy(out, bi, 32);
  for (int i = 2; i <= 8; i++) {
    for (int j = 0; j < 32; j++) cur[j] = b0[j] ^ bi[j];
    cur[32] = (uint8_t)i;
    m3x_oracle_sha256(cur, 33 + dst_len + 1, bi);
    memcpy(out + 32 * (i - 1), bi, 32);
  }
}

/* 64-byte big-endian -> Fp (mod p), result in Montgomery form */
static void fp_from_be64_mod(fp_t *r, const uint8_t b[64]) {
  uint64_t lo[6] = {0}, hi[6] = {0};
  for (int i = 0; i < 6; i++)
    for (int j = 0; j < 8; j++)
      lo[5 - i] = (lo[5 - i] << 8) | b[16 + 8 * i + j];
  for (int i = 0; i < 2; i++)
    for (int j = 0; j < 8; j++)
      hi[1 - i] = (hi[1 - i] << 8) | b[8 * i + j];
  while (ge_p(lo)) sub_p(lo);
  /* hi*2^384 mod p = mont_mul(hi, R2) in standard form */
  fp_t hif, r2s, hi384;
  memcpy(hif.v, hi, 48);
  memcpy(r2s.v, BLS_R2, 48);
  fp_mul_(&hi384, &hif, &r2s); /* hi * R2 * R^-1 = hi * R mod p (standard) */
  uint64_t sum[6];
  unsigned __int128 c = 0;
  for (int i = 0; i < 6; i++) {
    c += (unsigned __int128)hi384.v[i] + lo[i];
    sum[i] = (uint64_t)c;
    c >>= 64;
  }
  if (c || ge_p(sum)) sub_p(sum);
  fp_from_std(r, sum);
}

static void sswu_g2(g2_aff_t *out, const fp2_t *u) {
  /* RFC 9380 simplified SWU on E2'; non-constant-time (verification only) */
  fp2_t zu2, tv, x1, gx1, y1, x, y;
  fp2_sqr(&zu2, u);
  fp2_mul_(&zu2, &zu2, &SSWU_Z_);
  fp2_sqr(&tv, &zu2);
  fp2_add(&tv, &tv, &zu2);
  if (fp2_is_zero(&tv)) {
    /* x1 = B / (Z*A) */
    fp2_t za;
    fp2_mul_(&za, &SSWU_Z_, &SSWU_A_);
    fp2_inv(&za, &za);
    fp2_mul_(&x1, &SSWU_B_, &za);
  } else {
    fp2_t inv_tv, one_plus;
    fp2_inv(&inv_tv, &tv);
    fp2_add(&one_plus, &F2_ONE_, &inv_tv);
    fp2_t nb_over_a;
    fp2_inv(&nb_over_a, &SSWU_A_);
    fp2_mul_(&nb_over_a, &nb_over_a, &SSWU_B_);
    fp2_neg(&nb_over_a, &nb_over_a);
    fp2_mul_(&x1, &nb_over_a, &one_plus);
  }
  fp2_t ax, t;
  fp2_sqr(&gx1, &x1);
  fp2_mul_(&gx1, &gx1, &x1);
  fp2_mul_(&ax, &SSWU_A_, &x1);
  fp2_add(&gx1, &gx1, &ax);
  fp2_add(&gx1, &gx1, &SSWU_B_);
  if (fp2_sqrt(&y1, &gx1)) {
    x = x1;
    y = y1;
  } else {
    fp2_t x2, gx2;
    fp2_mul_(&x2, &zu2, &x1);
    fp2_sqr(&gx2, &x2);
    fp2_mul_(&gx2, &gx2, &x2);
    fp2_mul_(&t, &SSWU_A_, &x2);
    fp2_add(&gx2, &gx2, &t);
    fp2_add(&gx2, &gx2, &SSWU_B_);
    fp2_sqrt(&y1, &gx2); /* must succeed */
    x = x2;
    y = y1;
  }
  if (fp2_sgn0(u) != fp2_sgn0(&y)) fp2_neg(&y, &y);
  out->x = x;
  out->y = y;
  out->inf = 0;
}

static void iso_map_g2(g2_aff_t *out, const g2_aff_t *in) {
  /* 3-isogeny E2' -> E2 (RFC 9380 App. E.3) via Horner */
  fp2_t xn, xd, yn, yd, t;
  xn = ISO_KX[3];
  for (int i = 2; i >= 0; i--) {
    fp2_mul_(&xn, &xn, &in->x);
    fp2_add(&xn, &xn, &ISO_KX[i]);
  }
  xd = ISO_KXD[2];
  for (int i = 1; i >= 0; i--) {
    fp2_mul_(&xd, &xd, &in->x);
    fp2_add(&xd, &xd, &ISO_KXD[i]);
  }
  yn = ISO_KY[3];
  for (int i = 2; i >= 0; i--) {
    fp2_mul_(&yn, &yn, &in->x);
    fp2_add(&yn, &yn, &ISO_KY[i]);
  }
  yd = ISO_KYD[3];
  for (int i = 2; i >= 0; i--) {
    fp2_mul_(&yd, &yd, &in->x);
    fp2_add(&yd, &yd, &ISO_KYD[i]);
  }
  /* one shared inversion: 1/(xd*yd), then multiply back (3 extra muls
   * replace the second 381-bit inversion pow) */
  fp2_t prod, xdi, ydi;
  fp2_mul_(&prod, &xd, &yd);
  fp2_inv(&t, &prod);
  fp2_mul_(&ydi, &t, &xd); /* 1/yd */
  fp2_mul_(&xdi, &t, &yd); /* 1/xd */
  fp2_mul_(&out->x, &xn, &xdi);
  fp2_mul_(&out->y, &yn, &ydi);
  fp2_mul_(&out->y, &out->y, &in->y);
  out->inf = 0;
}

static void psi_g2(g2_aff_t *r, const g2_aff_t *p) {
  if (p->inf) {
    *r = *p;
    return;
  }
  fp2_t t;
  fp2_conj(&t, &p->x);
  fp2_mul_(&r->x, &t, &PSI_CX_);
  fp2_conj(&t, &p->y);
  fp2_mul_(&r->y, &t, &PSI_CY_);
  r->inf = 0;
}

static void g2_mul_u64(g2_jac_t *r, const g2_aff_t *p, uint64_t k) {
  uint8_t be[8];
  for (int i = 0; i < 8; i++) be[i] = (uint8_t)(k >> (56 - 8 * i));
  g2_mul_be(r, p, be, 8);
}

static void g2_jac_neg(g2_jac_t *r, const g2_jac_t *p) {
  r->x = p->x;
  fp2_neg(&r->y, &p->y);
  r->z = p->z;
}

/* psi on Jacobian coordinates: (cx*conj(X), cy*conj(Y), conj(Z)) — no
 * inversion (X'/Z'^2 = cx*conj(X/Z^2), Y'/Z'^3 = cy*conj(Y/Z^3)) */
static void psi_g2_jac(g2_jac_t *r, const g2_jac_t *p) {
  fp2_t t;
  fp2_conj(&t, &p->x);
  fp2_mul_(&r->x, &t, &PSI_CX_);
  fp2_conj(&t, &p->y);
  fp2_mul_(&r->y, &t, &PSI_CY_);
  fp2_conj(&r->z, &p->z);
}

static void g2_mul_u64_jac(g2_jac_t *r, const g2_jac_t *base, uint64_t k) {
  g2_jac_t acc;
  memset(&acc, 0, sizeof(acc));
  for (int b = 63; b >= 0; b--) {
    g2_dbl(&acc, &acc);
    if ((k >> b) & 1) g2_addj(&acc, &acc, base);
  }
  *r = acc;
}

static void clear_cofactor_g2_jac(g2_jac_t *out, const g2_jac_t *p) {
  /* Budroni-Pintore: [x^2-x-1]P + [x-1]psi(P) + psi^2([2]P); x negative.
   * Equals the RFC 9380 h_eff multiplication (asserted by the generator).
   * Fully Jacobian — no per-stage affine conversions (round-2 speedup;
   * results unchanged, pinned by the existing h2c fixtures). */
  g2_jac_t xp, xxp, acc, tmp, d;
  g2_mul_u64_jac(&xp, p, BLS_X_ABS);
  g2_jac_neg(&xp, &xp); /* [x]P */
  g2_mul_u64_jac(&xxp, &xp, BLS_X_ABS);
  g2_jac_neg(&xxp, &xxp); /* [x^2]P */
  acc = xxp;
  g2_jac_neg(&tmp, &xp);
  g2_addj(&acc, &acc, &tmp);
  g2_jac_neg(&tmp, p);
  g2_addj(&acc, &acc, &tmp); /* [x^2-x-1]P */
  d = xp;
  g2_jac_neg(&tmp, p);
  g2_addj(&d, &d, &tmp);
  psi_g2_jac(&tmp, &d);
  g2_addj(&acc, &acc, &tmp); /* + [x-1]psi(P) */
  g2_dbl(&tmp, p);
  psi_g2_jac(&tmp, &tmp);
  psi_g2_jac(&tmp, &tmp);
  g2_addj(&acc, &acc, &tmp); /* + psi^2([2]P) */
  *out = acc;
}

void h2c_g2(g2_aff_t *r, const uint8_t msg[32]) {
  uint8_t uni[256];
  expand_message_xmd(msg, 32, uni);
  fp2_t u0, u1;
  fp_from_be64_mod(&u0.c0, uni);
  fp_from_be64_mod(&u0.c1, uni + 64);
  fp_from_be64_mod(&u1.c0, uni + 128);
  fp_from_be64_mod(&u1.c1, uni + 192);
  g2_aff_t q0p, q1p, q0, q1;
  sswu_g2(&q0p, &u0);
  sswu_g2(&q1p, &u1);
  iso_map_g2(&q0, &q0p);
  iso_map_g2(&q1, &q1p);
  g2_jac_t s, t;
  g2_from_aff(&s, &q0);
  g2_from_aff(&t, &q1);
  g2_addj(&s, &s, &t);
  g2_jac_t cleared;
  clear_cofactor_g2_jac(&cleared, &s);
  g2_to_aff(r, &cleared); /* single affine conversion */
}

/* ---- RFC 9380 general forms (arbitrary DST / msg len; external-vector
 * pinning — the hot path keeps the fixed-DST fast forms above) ---- */

void m3x_oracle_expand_xmd(const uint8_t *msg, uint32_t msg_len,
                           const uint8_t *dst, uint32_t dst_len,
                           uint32_t len_in_bytes, uint8_t *out) {
  /* RFC 9380 §5.3.1, SHA-256 (b=32, s=64). Caller guarantees
   * len_in_bytes <= 8160 and dst_len <= 255. */
  uint32_t ell = (len_in_bytes + 31) / 32;
  uint8_t buf[64 + 1024 + 2 + 1 + 255 + 1];
  uint32_t off = 0;
  memset(buf, 0, 64);
  off = 64;
  memcpy(buf + off, msg, msg_len);
  off += msg_len;
  buf[off++] = (uint8_t)(len_in_bytes >> 8);
  buf[off++] = (uint8_t)len_in_bytes;
  buf[off++] = 0;
  memcpy(buf + off, dst, dst_len);
  off += dst_len;
  buf[off++] = (uint8_t)dst_len;
  uint8_t b0[32];
  m3x_oracle_sha256(buf, off, b0);
  uint8_t cur[32 + 1 + 255 + 1];
  memcpy(cur, b0, 32);
  cur[32] = 1;
  memcpy(cur + 33, dst, dst_len);
  cur[33 + dst_len] = (uint8_t)dst_len;
  uint8_t bi[32];
  m3x_oracle_sha256(cur, 33 + dst_len + 1, bi);
  uint32_t copied = len_in_bytes < 32 ? len_in_bytes : 32;
  memcpy(out, bi, copied);
  for (uint32_t blk = 2; blk <= ell; blk++) {
    for (int j = 0; j < 32; j++) cur[j] = b0[j] ^ bi[j];
    cur[32] = (uint8_t)blk;
    m3x_oracle_sha256(cur, 33 + dst_len + 1, bi);
    uint32_t base = 32 * (blk - 1);
    uint32_t nc = len_in_bytes - base < 32 ? len_in_bytes - base : 32;
    memcpy(out + base, bi, nc);
  }
}

int m3x_oracle_h2c_g2_dst(const uint8_t *msg, uint32_t msg_len,
                          const uint8_t *dst, uint32_t dst_len,
                          uint8_t out_uncomp[192]) {
  /* hash_to_curve for BLS12381G2_XMD:SHA-256_SSWU_RO_ with an arbitrary
   * DST (RFC 9380 §8.8.2) — same sswu/iso/cofactor code as h2c_g2. */
  bls_init();
  uint8_t uni[256];
  m3x_oracle_expand_xmd(msg, msg_len, dst, dst_len, 256, uni);
  fp2_t u0, u1;
  fp_from_be64_mod(&u0.c0, uni);
  fp_from_be64_mod(&u0.c1, uni + 64);
  fp_from_be64_mod(&u1.c0, uni + 128);
  fp_from_be64_mod(&u1.c1, uni + 192);
  g2_aff_t q0p, q1p, q0, q1;
  sswu_g2(&q0p, &u0);
  sswu_g2(&q1p, &u1);
  iso_map_g2(&q0, &q0p);
  iso_map_g2(&q1, &q1p);
  g2_jac_t sj, tj;
  g2_from_aff(&sj, &q0);
  g2_from_aff(&tj, &q1);
  g2_addj(&sj, &sj, &tj);
  g2_jac_t cleared;
  clear_cofactor_g2_jac(&cleared, &sj);
  g2_aff_t r;
  g2_to_aff(&r, &cleared);
  g2_to_uncomp(&r, out_uncomp);
  return 0;
}

/* map_to_curve WITHOUT cofactor clearing: yields an on-curve E'(Fp2)
 * point outside G2 w.h.p. — negative-subgroup test material */
int m3x_oracle_map_to_curve_g2_nococlear(const uint8_t msg[32],
                                         uint8_t out_uncomp[192]) {
  bls_init();
  uint8_t uni[256];
  expand_message_xmd(msg, 32, uni);
  fp2_t u0, u1;
  fp_from_be64_mod(&u0.c0, uni);
  fp_from_be64_mod(&u0.c1, uni + 64);
  fp_from_be64_mod(&u1.c0, uni + 128);
  fp_from_be64_mod(&u1.c1, uni + 192);
  g2_aff_t q0p, q1p, q0, q1;
  sswu_g2(&q0p, &u0);
  sswu_g2(&q1p, &u1);
  iso_map_g2(&q0, &q0p);
  iso_map_g2(&q1, &q1p);
  g2_jac_t sj, tj;
  g2_from_aff(&sj, &q0);
  g2_from_aff(&tj, &q1);
  g2_addj(&sj, &sj, &tj);
  g2_aff_t r;
  g2_to_aff(&r, &sj);
  g2_to_uncomp(&r, out_uncomp);
  return 0;
}

/* ----------------------------------------------------------------- init --- */

static void pp2_init(void) {
  /* PP2_ = 2 * p^2 as a 768-bit integer (lazy-reduction offset) */
  unsigned __int128 acc;
  uint64_t tmp[13] = {0};
  for (int i = 0; i < 6; i++) {
    uint64_t carry = 0;
    for (int j = 0; j < 6; j++) {
      acc = (unsigned __int128)BLS_P[i] * BLS_P[j] + tmp[i + j] + carry;
      tmp[i + j] = (uint64_t)acc;
      carry = (uint64_t)(acc >> 64);
    }
    tmp[i + 6] += carry;
  }
  uint64_t c = 0;
  for (int i = 0; i < 12; i++) {
    uint64_t v = (tmp[i] << 1) | c;
    c = tmp[i] >> 63;
    PP2_[i] = v;
  }
}

static int bls_init_done = 0;

void bls_init(void) {
  if (bls_init_done) return;
  pp2_init();
  /* Montgomery constants */
  memset(&FP_ZERO_, 0, sizeof(FP_ZERO_));
  memcpy(FP_R2_.v, BLS_R2, 48);
  uint64_t one_std[6] = {1, 0, 0, 0, 0, 0};
  fp_from_std(&FP_ONE_, one_std);
  /* (p-1)/2, p-2, (p+1)/4 in standard limbs */
  uint64_t t[6];
  memcpy(t, BLS_P, 48);
  t[0] -= 1; /* p odd, no borrow */
  for (int i = 0; i < 6; i++)
    P_HALF[i] = (t[i] >> 1) | (i < 5 ? (t[i + 1] << 63) : 0);
  memcpy(EXP_PM2, BLS_P, 48);
  EXP_PM2[0] -= 2;
  /* (p+1)/4: p+1 carries out of limb 0? p[0] = ...aaab, +1 no overflow */
  memcpy(t, BLS_P, 48);
  t[0] += 1;
  for (int i = 0; i < 6; i++)
    EXP_SQRT[i] = (t[i] >> 2) | (i < 5 ? (t[i + 1] << 62) : 0);
  /* field constants */
  memset(&F2_ZERO_, 0, sizeof(F2_ZERO_));
  memset(&F2_ONE_, 0, sizeof(F2_ONE_));
  F2_ONE_.c0 = FP_ONE_;
  XI_.c0 = FP_ONE_;
  XI_.c1 = FP_ONE_;
  fp2_inv(&XI_INV_, &XI_);
  uint64_t two_std[6] = {2, 0, 0, 0, 0, 0};
  fp_t two;
  fp_from_std(&two, two_std);
  fp_inv(&TWO_INV_.c0, &two);
  memset(&TWO_INV_.c1, 0, sizeof(fp_t));
  uint64_t four_std[6] = {4, 0, 0, 0, 0, 0};
  fp_from_std(&B1_, four_std);
  fp2_t four2;
  memset(&four2, 0, sizeof(four2));
  fp_from_std(&four2.c0, four_std);
  fp2_mul_(&B2_, &four2, &XI_);
#define LOAD_FP2(dst, name)                                                   \
  do {                                                                        \
    fp_from_std(&(dst).c0, name##_C0);                                        \
    fp_from_std(&(dst).c1, name##_C1);                                        \
  } while (0)
  LOAD_FP2(SSWU_A_, SSWU_A);
  LOAD_FP2(SSWU_B_, SSWU_B);
  LOAD_FP2(SSWU_Z_, SSWU_Z);
  LOAD_FP2(PSI_CX_, PSI_CX);
  LOAD_FP2(PSI_CY_, PSI_CY);
  LOAD_FP2(ISO_KX[0], ISO_XNUM0);
  LOAD_FP2(ISO_KX[1], ISO_XNUM1);
  LOAD_FP2(ISO_KX[2], ISO_XNUM2);
  LOAD_FP2(ISO_KX[3], ISO_XNUM3);
  LOAD_FP2(ISO_KXD[0], ISO_XDEN0);
  LOAD_FP2(ISO_KXD[1], ISO_XDEN1);
  LOAD_FP2(ISO_KXD[2], ISO_XDEN2);
  LOAD_FP2(ISO_KY[0], ISO_YNUM0);
  LOAD_FP2(ISO_KY[1], ISO_YNUM1);
  LOAD_FP2(ISO_KY[2], ISO_YNUM2);
  LOAD_FP2(ISO_KY[3], ISO_YNUM3);
  LOAD_FP2(ISO_KYD[0], ISO_YDEN0);
  LOAD_FP2(ISO_KYD[1], ISO_YDEN1);
  LOAD_FP2(ISO_KYD[2], ISO_YDEN2);
  LOAD_FP2(ISO_KYD[3], ISO_YDEN3);
  fp2_t fw1, z6;
  LOAD_FP2(fw1, FROB_W1);
  LOAD_FP2(z6, ZETA6);
  FW1_POW[0] = F2_ONE_;
  Z6_POW[0] = F2_ONE_;
  for (int i = 1; i < 6; i++) {
    fp2_mul_(&FW1_POW[i], &FW1_POW[i - 1], &fw1);
    fp2_mul_(&Z6_POW[i], &Z6_POW[i - 1], &z6);
  }
  fp2_t fw2;
  {
    fp2_t c;
    fp2_conj(&c, &fw1);
    fp2_mul_(&fw2, &fw1, &c); /* FW1^(p+1) = norm(FW1) */
  }
  FW2_POW[0] = F2_ONE_;
  for (int i = 1; i < 6; i++) fp2_mul_(&FW2_POW[i], &FW2_POW[i - 1], &fw2);
  /* generators */
  fp_from_std(&G1_GEN.x, BLS_G1X);
  fp_from_std(&G1_GEN.y, BLS_G1Y);
  G1_GEN.inf = 0;
  LOAD_FP2(G2_GEN.x, BLS_G2X);
  LOAD_FP2(G2_GEN.y, BLS_G2Y);
  G2_GEN.inf = 0;
  /* order bytes (big-endian) */
  for (int i = 0; i < 4; i++)
    for (int j = 0; j < 8; j++)
      ORDER_BE[8 * i + j] = (uint8_t)(BLS_ORDER[3 - i] >> (56 - 8 * j));
  bls_init_done = 1;
}
