/* hbls_oracle.c — CPU oracle: C restatement of the herumi bls/mcl BLS12-381
 * semantics Harmony's consensus hot path uses (BLS_SWAP_G: pk in G1, sig in G2).
 *
 * TEST INFRASTRUCTURE + CPU BASELINE ONLY.  Only tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg may call this library; the product path is
 * the HIP C-ABI library under harmony_amd/ and must fail loudly without a GPU.
 *
 * Semantics restated from (see oracle/pyref.py for the full provenance notes):
 *   - harmony crypto/bls/bls.go:17-20 (sizes), mask.go:57-153 (aggregate/mask)
 *   - harmony-one/bls src/bls_c_impl.hpp: blsSignHash/blsVerifyHash/toG
 *   - harmony-one/mcl: Fp::setArrayMask, MapTo::calcBN (Fouque-Tibouchi),
 *     EcT::save/load IoSerialize (LE + parity flag), Fp2 squareRoot
 * Pinned by golden vectors: 26 sk->pk pairs (.hmy) incl. the BLS_SWAP_G base
 * point = [h1]*FTmap(1); 3420 genesis pubkeys (deserialize+subgroup).
 * Validated against oracle/pyref.py on every exported entry point.
 *
 * Build: gcc -O3 -fopenmp -shared -fPIC hbls_oracle.c -o libhbls_oracle.so
 */
#include <stdint.h>
#include <string.h>
#include <stddef.h>
#include "bls_consts.h"

#ifdef _OPENMP
#include <omp.h>
#endif

typedef unsigned __int128 u128;

/* ================================================================== Fp */
typedef struct { uint64_t l[6]; } fp_t;   /* Montgomery form, 6x64 LE limbs */
typedef struct { fp_t a, b; } fp2_t;      /* a + b*u, u^2 = -1 */

static const fp_t FP_ZERO = {{0, 0, 0, 0, 0, 0}};

static inline int fp_is_zero(const fp_t *x) {
    return (x->l[0] | x->l[1] | x->l[2] | x->l[3] | x->l[4] | x->l[5]) == 0;
}
static inline int fp_eq(const fp_t *x, const fp_t *y) {
    return memcmp(x->l, y->l, 48) == 0;
}
static inline int fp_geq_p(const uint64_t t[6]) {
    for (int i = 5; i >= 0; i--) {
        if (t[i] > BLS_P[i]) return 1;
        if (t[i] < BLS_P[i]) return 0;
    }
    return 1; /* equal */
}
static inline void fp_cond_sub_p(fp_t *r, const uint64_t t[6], uint64_t hi) {
    if (hi || fp_geq_p(t)) {
        u128 bw = 0;
        for (int i = 0; i < 6; i++) {
            u128 d = (u128)t[i] - BLS_P[i] - (uint64_t)bw;
            r->l[i] = (uint64_t)d;
            bw = (d >> 64) & 1; /* borrow */
        }
    } else {
        memcpy(r->l, t, 48);
    }
}
static void fp_add(fp_t *r, const fp_t *x, const fp_t *y) {
    uint64_t t[6]; u128 c = 0;
    for (int i = 0; i < 6; i++) {
        c += (u128)x->l[i] + y->l[i];
        t[i] = (uint64_t)c; c >>= 64;
    }
    fp_cond_sub_p(r, t, (uint64_t)c);
}
static void fp_sub(fp_t *r, const fp_t *x, const fp_t *y) {
    uint64_t t[6]; u128 bw = 0;
    for (int i = 0; i < 6; i++) {
        u128 d = (u128)x->l[i] - y->l[i] - (uint64_t)bw;
        t[i] = (uint64_t)d; bw = (d >> 64) & 1;
    }
    if (bw) { /* add p back */
        u128 c = 0;
        for (int i = 0; i < 6; i++) {
            c += (u128)t[i] + BLS_P[i];
            t[i] = (uint64_t)c; c >>= 64;
        }
    }
    memcpy(r->l, t, 48);
}
static void fp_neg(fp_t *r, const fp_t *x) {
    if (fp_is_zero(x)) { *r = *x; return; }
    uint64_t t[6]; u128 bw = 0;
    for (int i = 0; i < 6; i++) {
        u128 d = (u128)BLS_P[i] - x->l[i] - (uint64_t)bw;
        t[i] = (uint64_t)d; bw = (d >> 64) & 1;
    }
    memcpy(r->l, t, 48);
}
static void fp_dbl(fp_t *r, const fp_t *x) { fp_add(r, x, x); }

/* algorithmic-work counter: every Fp multiplication (the unit all cost
 * estimates in SURVEY.md §8d are stated in).  Thread-local; used by bench.py
 * to derive the EXACT algorithmic op count of one aggregate-verify. */
static __thread uint64_t g_fp_mul_count = 0;
void oracle_reset_op_count(void) { g_fp_mul_count = 0; }
uint64_t oracle_op_count(void) { return g_fp_mul_count; }

/* CIOS Montgomery multiplication */
static void fp_mul(fp_t *r, const fp_t *x, const fp_t *y) {
    g_fp_mul_count++;
    uint64_t t[8];
    memset(t, 0, sizeof(t));
    for (int i = 0; i < 6; i++) {
        u128 acc = 0;
        uint64_t xi = x->l[i];
        for (int j = 0; j < 6; j++) {
            acc = (u128)xi * y->l[j] + t[j] + (uint64_t)(acc >> 64);
            t[j] = (uint64_t)acc;
        }
        acc = (u128)t[6] + (uint64_t)(acc >> 64);
        t[6] = (uint64_t)acc;
        t[7] = (uint64_t)(acc >> 64);
        uint64_t m = t[0] * BLS_P_INV;
        acc = (u128)m * BLS_P[0] + t[0];
        for (int j = 1; j < 6; j++) {
            acc = (u128)m * BLS_P[j] + t[j] + (uint64_t)(acc >> 64);
            t[j - 1] = (uint64_t)acc;
        }
        acc = (u128)t[6] + (uint64_t)(acc >> 64);
        t[5] = (uint64_t)acc;
        t[6] = t[7] + (uint64_t)(acc >> 64);
        t[7] = 0;
    }
    fp_cond_sub_p(r, t, t[6]);
}
static void fp_sqr(fp_t *r, const fp_t *x) { fp_mul(r, x, x); }

static void fp_from_u64(fp_t *r, uint64_t v) {
    fp_t t = FP_ZERO; t.l[0] = v;
    fp_t r2; memcpy(r2.l, BLS_R2P, 48);
    fp_mul(r, &t, &r2);
}
static void fp_to_mont(fp_t *r, const uint64_t v[6]) {
    fp_t t; memcpy(t.l, v, 48);
    fp_t r2; memcpy(r2.l, BLS_R2P, 48);
    fp_mul(r, &t, &r2);
}
static void fp_from_mont(uint64_t r[6], const fp_t *x) {
    /* multiply by 1 (non-Montgomery) == Montgomery reduce */
    fp_t one_raw = FP_ZERO; one_raw.l[0] = 1;
    fp_t t;
    fp_mul(&t, x, &one_raw);
    memcpy(r, t.l, 48);
}
/* MSB-first square-and-multiply; exponent given as LE limbs (plain integer) */
static void fp_pow(fp_t *r, const fp_t *a, const uint64_t *e, int n) {
    fp_t acc; memcpy(acc.l, BLS_ONE_P, 48);
    int started = 0;
    for (int i = n - 1; i >= 0; i--) {
        for (int b = 63; b >= 0; b--) {
            if (started) fp_sqr(&acc, &acc);
            if ((e[i] >> b) & 1) {
                if (started) fp_mul(&acc, &acc, a);
                else { acc = *a; started = 1; }
            }
        }
    }
    *r = acc;
}
static void fp_inv(fp_t *r, const fp_t *x) { fp_pow(r, x, BLS_PM2, 6); }
/* sqrt = x^((p+1)/4); returns 1 iff root exists.  NO canonicalization
 * (mcl SquareRoot 3-mod-4 branch — pinned via the base-point derivation). */
static int fp_sqrt(fp_t *r, const fp_t *x) {
    fp_t y, y2;
    fp_pow(&y, x, BLS_SQRT_EXP, 6);
    fp_sqr(&y2, &y);
    if (!fp_eq(&y2, x)) return 0;
    *r = y;
    return 1;
}
/* legendre: 1 QR, -1 non-QR, 0 zero */
static int fp_legendre(const fp_t *x) {
    if (fp_is_zero(x)) return 0;
    fp_t y;
    fp_pow(&y, x, BLS_LEG_EXP, 6);
    fp_t one; memcpy(one.l, BLS_ONE_P, 48);
    return fp_eq(&y, &one) ? 1 : -1;
}
static int fp_is_odd(const fp_t *x) {
    uint64_t raw[6];
    fp_from_mont(raw, x);
    return raw[0] & 1;
}

/* ================================================================== Fp2 */
static inline int fp2_is_zero(const fp2_t *x) { return fp_is_zero(&x->a) && fp_is_zero(&x->b); }
static inline int fp2_eq(const fp2_t *x, const fp2_t *y) { return fp_eq(&x->a, &y->a) && fp_eq(&x->b, &y->b); }
static void fp2_add(fp2_t *r, const fp2_t *x, const fp2_t *y) { fp_add(&r->a, &x->a, &y->a); fp_add(&r->b, &x->b, &y->b); }
static void fp2_sub(fp2_t *r, const fp2_t *x, const fp2_t *y) { fp_sub(&r->a, &x->a, &y->a); fp_sub(&r->b, &x->b, &y->b); }
static void fp2_neg(fp2_t *r, const fp2_t *x) { fp_neg(&r->a, &x->a); fp_neg(&r->b, &x->b); }
static void fp2_conj(fp2_t *r, const fp2_t *x) { r->a = x->a; fp_neg(&r->b, &x->b); }
static void fp2_dbl(fp2_t *r, const fp2_t *x) { fp2_add(r, x, x); }
static void fp2_mul(fp2_t *r, const fp2_t *x, const fp2_t *y) {
    fp_t ac, bd, ab, cd, t;
    fp_mul(&ac, &x->a, &y->a);
    fp_mul(&bd, &x->b, &y->b);
    fp_add(&ab, &x->a, &x->b);
    fp_add(&cd, &y->a, &y->b);
    fp_mul(&t, &ab, &cd);
    fp_sub(&t, &t, &ac);
    fp_sub(&t, &t, &bd);
    fp_sub(&r->a, &ac, &bd);
    r->b = t;
}
static void fp2_sqr(fp2_t *r, const fp2_t *x) {
    fp_t s, d, m;
    fp_add(&s, &x->a, &x->b);
    fp_sub(&d, &x->a, &x->b);
    fp_mul(&m, &x->a, &x->b);
    fp_mul(&r->a, &s, &d);
    fp_dbl(&r->b, &m);
}
static void fp2_mul_fp(fp2_t *r, const fp2_t *x, const fp_t *s) {
    fp_mul(&r->a, &x->a, s);
    fp_mul(&r->b, &x->b, s);
}
static void fp2_mul_xi(fp2_t *r, const fp2_t *x) { /* * (1+u) */
    fp_t na, nb;
    fp_sub(&na, &x->a, &x->b);
    fp_add(&nb, &x->a, &x->b);
    r->a = na; r->b = nb;
}
static void fp2_inv(fp2_t *r, const fp2_t *x) {
    fp_t n, t, ia, ib;
    fp_sqr(&n, &x->a);
    fp_sqr(&t, &x->b);
    fp_add(&n, &n, &t);
    fp_inv(&n, &n);
    fp_mul(&ia, &x->a, &n);
    fp_mul(&t, &x->b, &n);
    fp_neg(&ib, &t);
    r->a = ia; r->b = ib;
}
/* mcl Fp2T::squareRoot restatement (root choice follows fp_sqrt) */
static int fp2_sqrt(fp2_t *r, const fp2_t *x) {
    if (fp_is_zero(&x->b)) {
        fp_t s;
        if (fp_sqrt(&s, &x->a)) { r->a = s; r->b = FP_ZERO; return 1; }
        fp_t na; fp_neg(&na, &x->a);
        if (fp_sqrt(&s, &na)) { r->a = FP_ZERO; r->b = s; return 1; }
        return 0;
    }
    fp_t n, t, w, c, d, inv2c;
    fp_sqr(&n, &x->a);
    fp_sqr(&t, &x->b);
    fp_add(&n, &n, &t);          /* norm = a^2+b^2 */
    if (!fp_sqrt(&w, &n)) return 0;
    /* t = (a+w)/2, else (a-w)/2 */
    extern const fp_t *hbls_inv2(void);
    const fp_t *inv2 = hbls_inv2();
    fp_add(&t, &x->a, &w);
    fp_mul(&t, &t, inv2);
    if (!fp_sqrt(&c, &t)) {
        fp_sub(&t, &x->a, &w);
        fp_mul(&t, &t, inv2);
        if (!fp_sqrt(&c, &t)) return 0;
    }
    fp_dbl(&inv2c, &c);
    fp_inv(&inv2c, &inv2c);
    fp_mul(&d, &x->b, &inv2c);
    r->a = c; r->b = d;
    return 1;
}
static int fp2_is_odd(const fp2_t *x) { return fp_is_odd(&x->a); }

/* library-init cached constants (thread-safe: set once before any threads) */
static fp_t G_INV2;
const fp_t *hbls_inv2(void) { return &G_INV2; }
__attribute__((constructor)) static void hbls_oracle_init(void) {
    fp_t two;
    fp_from_u64(&two, 2);
    fp_inv(&G_INV2, &two);
}

/* ================================================================== G1 (Jacobian) */
typedef struct { fp_t x, y, z; } g1_t;   /* z==0 => infinity */
typedef struct { fp_t x, y; } g1aff_t;

static void g1_set_inf(g1_t *p) { memcpy(p->x.l, BLS_ONE_P, 48); memcpy(p->y.l, BLS_ONE_P, 48); p->z = FP_ZERO; }
static int g1_is_inf(const g1_t *p) { return fp_is_zero(&p->z); }

static void g1_dbl(g1_t *r, const g1_t *p) {
    if (g1_is_inf(p)) { *r = *p; return; }
    fp_t A, B, C, D, E, F, t;
    fp_sqr(&A, &p->x);
    fp_sqr(&B, &p->y);
    fp_sqr(&C, &B);
    fp_add(&t, &p->x, &B);
    fp_sqr(&t, &t);
    fp_sub(&t, &t, &A);
    fp_sub(&t, &t, &C);
    fp_dbl(&D, &t);              /* 4XB */
    fp_dbl(&E, &A);
    fp_add(&E, &E, &A);          /* 3A */
    fp_sqr(&F, &E);
    fp_sub(&r->x, &F, &D);
    fp_sub(&r->x, &r->x, &D);    /* F - 2D */
    fp_mul(&t, &p->y, &p->z);
    fp_dbl(&r->z, &t);           /* 2YZ */
    fp_sub(&t, &D, &r->x);
    fp_mul(&t, &E, &t);
    fp_dbl(&C, &C); fp_dbl(&C, &C); fp_dbl(&C, &C);  /* 8C */
    fp_sub(&r->y, &t, &C);
}
static void g1_add(g1_t *r, const g1_t *p, const g1_t *q) {
    if (g1_is_inf(p)) { *r = *q; return; }
    if (g1_is_inf(q)) { *r = *p; return; }
    fp_t z1z1, z2z2, u1, u2, s1, s2, h, rr, hh, hhh, v, t;
    fp_sqr(&z1z1, &p->z);
    fp_sqr(&z2z2, &q->z);
    fp_mul(&u1, &p->x, &z2z2);
    fp_mul(&u2, &q->x, &z1z1);
    fp_mul(&s1, &p->y, &q->z); fp_mul(&s1, &s1, &z2z2);
    fp_mul(&s2, &q->y, &p->z); fp_mul(&s2, &s2, &z1z1);
    fp_sub(&h, &u2, &u1);
    fp_sub(&rr, &s2, &s1);
    if (fp_is_zero(&h)) {
        if (fp_is_zero(&rr)) { g1_dbl(r, p); return; }
        g1_set_inf(r); return;
    }
    fp_sqr(&hh, &h);
    fp_mul(&hhh, &hh, &h);
    fp_mul(&v, &u1, &hh);
    fp_sqr(&t, &rr);
    fp_sub(&t, &t, &hhh);
    fp_sub(&t, &t, &v);
    fp_sub(&r->x, &t, &v);
    fp_sub(&t, &v, &r->x);
    fp_mul(&t, &rr, &t);
    fp_mul(&v, &s1, &hhh);
    fp_sub(&r->y, &t, &v);
    fp_mul(&t, &p->z, &q->z);
    fp_mul(&r->z, &t, &h);
}
static void g1_neg(g1_t *r, const g1_t *p) { r->x = p->x; fp_neg(&r->y, &p->y); r->z = p->z; }
static void g1_to_affine(g1aff_t *r, const g1_t *p) {
    fp_t zi, zi2, zi3;
    fp_inv(&zi, &p->z);
    fp_sqr(&zi2, &zi);
    fp_mul(&zi3, &zi2, &zi);
    fp_mul(&r->x, &p->x, &zi2);
    fp_mul(&r->y, &p->y, &zi3);
}
static void g1_from_affine(g1_t *r, const g1aff_t *p) {
    r->x = p->x; r->y = p->y; memcpy(r->z.l, BLS_ONE_P, 48);
}
/* scalar mult, scalar = LE limbs plain integer */
static void g1_mul(g1_t *r, const g1_t *p, const uint64_t *k, int n) {
    g1_t acc; g1_set_inf(&acc);
    int started = 0;
    for (int i = n - 1; i >= 0; i--)
        for (int b = 63; b >= 0; b--) {
            if (started) g1_dbl(&acc, &acc);
            if ((k[i] >> b) & 1) { g1_add(&acc, &acc, p); started = 1; }
        }
    *r = acc;
}

/* ================================================================== G2 (Jacobian) */
typedef struct { fp2_t x, y, z; } g2_t;
typedef struct { fp2_t x, y; } g2aff_t;
static const fp2_t FP2_ZERO = {{{0,0,0,0,0,0}}, {{0,0,0,0,0,0}}};

static void fp2_one(fp2_t *r) { memcpy(r->a.l, BLS_ONE_P, 48); r->b = FP_ZERO; }
static void g2_set_inf(g2_t *p) { fp2_one(&p->x); fp2_one(&p->y); p->z = FP2_ZERO; }
static int g2_is_inf(const g2_t *p) { return fp2_is_zero(&p->z); }

static void g2_dbl(g2_t *r, const g2_t *p) {
    if (g2_is_inf(p)) { *r = *p; return; }
    fp2_t A, B, C, D, E, F, t;
    fp2_sqr(&A, &p->x);
    fp2_sqr(&B, &p->y);
    fp2_sqr(&C, &B);
    fp2_add(&t, &p->x, &B);
    fp2_sqr(&t, &t);
    fp2_sub(&t, &t, &A);
    fp2_sub(&t, &t, &C);
    fp2_dbl(&D, &t);
    fp2_dbl(&E, &A);
    fp2_add(&E, &E, &A);
    fp2_sqr(&F, &E);
    fp2_sub(&r->x, &F, &D);
    fp2_sub(&r->x, &r->x, &D);
    fp2_mul(&t, &p->y, &p->z);
    fp2_dbl(&r->z, &t);
    fp2_sub(&t, &D, &r->x);
    fp2_mul(&t, &E, &t);
    fp2_dbl(&C, &C); fp2_dbl(&C, &C); fp2_dbl(&C, &C);
    fp2_sub(&r->y, &t, &C);
}
static void g2_add(g2_t *r, const g2_t *p, const g2_t *q) {
    if (g2_is_inf(p)) { *r = *q; return; }
    if (g2_is_inf(q)) { *r = *p; return; }
    fp2_t z1z1, z2z2, u1, u2, s1, s2, h, rr, hh, hhh, v, t;
    fp2_sqr(&z1z1, &p->z);
    fp2_sqr(&z2z2, &q->z);
    fp2_mul(&u1, &p->x, &z2z2);
    fp2_mul(&u2, &q->x, &z1z1);
    fp2_mul(&s1, &p->y, &q->z); fp2_mul(&s1, &s1, &z2z2);
    fp2_mul(&s2, &q->y, &p->z); fp2_mul(&s2, &s2, &z1z1);
    fp2_sub(&h, &u2, &u1);
    fp2_sub(&rr, &s2, &s1);
    if (fp2_is_zero(&h)) {
        if (fp2_is_zero(&rr)) { g2_dbl(r, p); return; }
        g2_set_inf(r); return;
    }
    fp2_sqr(&hh, &h);
    fp2_mul(&hhh, &hh, &h);
    fp2_mul(&v, &u1, &hh);
    fp2_sqr(&t, &rr);
    fp2_sub(&t, &t, &hhh);
    fp2_sub(&t, &t, &v);
    fp2_sub(&r->x, &t, &v);
    fp2_sub(&t, &v, &r->x);
    fp2_mul(&t, &rr, &t);
    fp2_mul(&v, &s1, &hhh);
    fp2_sub(&r->y, &t, &v);
    fp2_mul(&t, &p->z, &q->z);
    fp2_mul(&r->z, &t, &h);
}
static void g2_neg(g2_t *r, const g2_t *p) { r->x = p->x; fp2_neg(&r->y, &p->y); r->z = p->z; }
static void g2_to_affine(g2aff_t *r, const g2_t *p) {
    fp2_t zi, zi2, zi3;
    fp2_inv(&zi, &p->z);
    fp2_sqr(&zi2, &zi);
    fp2_mul(&zi3, &zi2, &zi);
    fp2_mul(&r->x, &p->x, &zi2);
    fp2_mul(&r->y, &p->y, &zi3);
}
static void g2_from_affine(g2_t *r, const g2aff_t *p) {
    r->x = p->x; r->y = p->y; fp2_one(&r->z);
}
static void g2_mul(g2_t *r, const g2_t *p, const uint64_t *k, int n) {
    g2_t acc; g2_set_inf(&acc);
    int started = 0;
    for (int i = n - 1; i >= 0; i--)
        for (int b = 63; b >= 0; b--) {
            if (started) g2_dbl(&acc, &acc);
            if ((k[i] >> b) & 1) { g2_add(&acc, &acc, p); started = 1; }
        }
    *r = acc;
}
static int g2_eq(const g2_t *p, const g2_t *q) {
    if (g2_is_inf(p) || g2_is_inf(q)) return g2_is_inf(p) && g2_is_inf(q);
    fp2_t z1z1, z2z2, a, b;
    fp2_sqr(&z1z1, &p->z); fp2_sqr(&z2z2, &q->z);
    fp2_mul(&a, &p->x, &z2z2); fp2_mul(&b, &q->x, &z1z1);
    if (!fp2_eq(&a, &b)) return 0;
    fp2_mul(&a, &p->y, &q->z); fp2_mul(&a, &a, &z2z2);
    fp2_mul(&b, &q->y, &p->z); fp2_mul(&b, &b, &z1z1);
    return fp2_eq(&a, &b);
}
/* psi endomorphism: (x,y) -> (cx*conj(x), cy*conj(y)) */
static void g2_psi_jac(g2_t *r, const g2_t *p) {
    /* x=X/Z^2, y=Y/Z^3 -> (cx*conj(X), cy*conj(Y), conj(Z)): no inversion */
    if (g2_is_inf(p)) { *r = *p; return; }
    fp2_t cx, cy, t;
    memcpy(cx.a.l, BLS_PSI_CX_A, 48); memcpy(cx.b.l, BLS_PSI_CX_B, 48);
    memcpy(cy.a.l, BLS_PSI_CY_A, 48); memcpy(cy.b.l, BLS_PSI_CY_B, 48);
    fp2_conj(&t, &p->x); fp2_mul(&r->x, &t, &cx);
    fp2_conj(&t, &p->y); fp2_mul(&r->y, &t, &cy);
    fp2_conj(&r->z, &p->z);
}
static void g2_psi(g2_t *r, const g2_t *p) { g2_psi_jac(r, p); }

/* ================================================================== serialization */
static void fp_to_le48(uint8_t out[48], const fp_t *x) {
    uint64_t raw[6];
    fp_from_mont(raw, x);
    for (int i = 0; i < 6; i++)
        for (int j = 0; j < 8; j++)
            out[i * 8 + j] = (uint8_t)(raw[i] >> (8 * j));
}
static int fp_from_le48(fp_t *x, const uint8_t in[48]) {
    uint64_t raw[6];
    for (int i = 0; i < 6; i++) {
        raw[i] = 0;
        for (int j = 0; j < 8; j++)
            raw[i] |= (uint64_t)in[i * 8 + j] << (8 * j);
    }
    if (fp_geq_p(raw)) return 0;
    fp_to_mont(x, raw);
    return 1;
}
static int is_all_zero(const uint8_t *b, size_t n) {
    for (size_t i = 0; i < n; i++) if (b[i]) return 0;
    return 1;
}
void oracle_g1_serialize(uint8_t out[48], const g1_t *p) {
    if (g1_is_inf(p)) { memset(out, 0, 48); return; }
    g1aff_t a;
    g1_to_affine(&a, p);
    fp_to_le48(out, &a.x);
    if (fp_is_odd(&a.y)) out[47] |= 0x80;
}
void oracle_g2_serialize(uint8_t out[96], const g2_t *p) {
    if (g2_is_inf(p)) { memset(out, 0, 96); return; }
    g2aff_t a;
    g2_to_affine(&a, p);
    fp_to_le48(out, &a.x.a);
    fp_to_le48(out + 48, &a.x.b);
    if (fp2_is_odd(&a.y)) out[95] |= 0x80;
}

/* subgroup checks: multiply by r (exact membership — method-independent) */
static int g1_in_subgroup(const g1_t *p) {
    g1_t t;
    g1_mul(&t, p, BLS_R, 4);
    return g1_is_inf(&t);
}
static int g2_in_subgroup(const g2_t *p) {
    g2_t t;
    g2_mul(&t, p, BLS_R, 4);
    return g2_is_inf(&t);
}
static void g2_psi_jac(g2_t *r, const g2_t *p);
static int g2_eq(const g2_t *p, const g2_t *q);
/* Scott's criterion: Q in G2 iff psi(Q) == [z]Q (z<0) — same accept set as
 * the [r]Q test (equivalence property-tested in tests/test_oracle.py) */
static int g2_in_subgroup_fast(const g2_t *p) {
    if (g2_is_inf(p)) return 1;
    g2_t lhs, rhs;
    g2_psi_jac(&lhs, p);
    uint64_t u = BLS_U;
    g2_mul(&rhs, p, &u, 1);
    g2_neg(&rhs, &rhs);
    return g2_eq(&lhs, &rhs);
}

static void fp2_b2(fp2_t *b) { memcpy(b->a.l, BLS_B2_A, 48); memcpy(b->b.l, BLS_B2_B, 48); }

int oracle_g1_deserialize(g1_t *p, const uint8_t in[48], int check_subgroup) {
    if (is_all_zero(in, 48)) { g1_set_inf(p); return 1; }
    uint8_t buf[48];
    memcpy(buf, in, 48);
    int odd = (buf[47] & 0x80) != 0;
    buf[47] &= 0x7F;
    g1aff_t a;
    if (!fp_from_le48(&a.x, buf)) return 0;
    fp_t y2, b1;
    fp_sqr(&y2, &a.x); fp_mul(&y2, &y2, &a.x);
    memcpy(b1.l, BLS_B1, 48);
    fp_add(&y2, &y2, &b1);
    if (!fp_sqrt(&a.y, &y2)) return 0;
    if (fp_is_odd(&a.y) != odd) fp_neg(&a.y, &a.y);
    g1_from_affine(p, &a);
    if (check_subgroup && !g1_in_subgroup(p)) return 0;
    return 1;
}
int oracle_g2_deserialize(g2_t *p, const uint8_t in[96], int check_subgroup) {
    if (is_all_zero(in, 96)) { g2_set_inf(p); return 1; }
    uint8_t buf[96];
    memcpy(buf, in, 96);
    int odd = (buf[95] & 0x80) != 0;
    buf[95] &= 0x7F;
    g2aff_t a;
    if (!fp_from_le48(&a.x.a, buf)) return 0;
    if (!fp_from_le48(&a.x.b, buf + 48)) return 0;
    fp2_t y2, b2;
    fp2_sqr(&y2, &a.x); fp2_mul(&y2, &y2, &a.x);
    fp2_b2(&b2);
    fp2_add(&y2, &y2, &b2);
    if (!fp2_sqrt(&a.y, &y2)) return 0;
    if (fp2_is_odd(&a.y) != odd) fp2_neg(&a.y, &a.y);
    g2_from_affine(p, &a);
    if (check_subgroup && !g2_in_subgroup(p)) return 0;
    return 1;
}

/* ================================================================== FT map (mcl calcBN) */
static void fp_base_point(g1_t *g) {
    g1aff_t a;
    memcpy(a.x.l, BLS_G1_X, 48);
    memcpy(a.y.l, BLS_G1_Y, 48);
    g1_from_affine(g, &a);
}
/* Fouque-Tibouchi over Fp2 (used for hash-to-G2); t in Fp2, never negative
 * for subfield inputs (norm-legendre); see pyref.ft_map_g2 */
static int ft_map_g2(g2aff_t *r, const fp2_t *t) {
    if (fp2_is_zero(t)) return 0;
    fp_t norm, tmp;
    fp_sqr(&norm, &t->a);
    fp_sqr(&tmp, &t->b);
    fp_add(&norm, &norm, &tmp);
    int neg = fp_legendre(&norm) < 0;
    fp2_t w, b2, x, y2, y;
    fp2_b2(&b2);
    fp2_sqr(&w, t);
    fp2_add(&w, &w, &b2);
    { fp_t one; memcpy(one.l, BLS_ONE_P, 48); fp_add(&w.a, &w.a, &one); }
    if (fp2_is_zero(&w)) return 0;
    fp2_inv(&w, &w);
    fp2_mul(&w, &w, t);
    { fp_t c1; memcpy(c1.l, BLS_FT_C1, 48); fp2_mul_fp(&w, &w, &c1); }
    for (int i = 0; i < 3; i++) {
        if (i == 0) {
            fp2_mul(&x, t, &w);
            fp2_neg(&x, &x);
            fp_t c2; memcpy(c2.l, BLS_FT_C2, 48);
            fp_add(&x.a, &x.a, &c2);
        } else if (i == 1) {
            fp2_neg(&x, &x);
            fp_t one; memcpy(one.l, BLS_ONE_P, 48);
            fp_sub(&x.a, &x.a, &one);
        } else {
            fp2_sqr(&x, &w);
            fp2_inv(&x, &x);
            fp_t one; memcpy(one.l, BLS_ONE_P, 48);
            fp_add(&x.a, &x.a, &one);
        }
        fp2_sqr(&y2, &x); fp2_mul(&y2, &y2, &x);
        fp2_add(&y2, &y2, &b2);
        if (fp2_sqrt(&y, &y2)) {
            if (neg) fp2_neg(&y, &y);
            r->x = x; r->y = y;
            return 1;
        }
    }
    return 0;
}

/* G2 cofactor clearing */
static int g_use_fast_cofactor = 1;
void oracle_set_g2_cofactor_mode(int fast) { g_use_fast_cofactor = fast; }

static void g2_mul_u64(g2_t *r, const g2_t *p, uint64_t k) { g2_mul(r, p, &k, 1); }

static void g2_clear_cofactor(g2_t *r, const g2_t *p) {
    if (!g_use_fast_cofactor) {
        g2_mul(r, p, BLS_H2, BLS_H2_LIMBS);
        return;
    }
    /* Budroni-Pintore: (z^2-z-1)P + (z-1)psi(P) + psi^2(2P), z = -BLS_U */
    g2_t t1, t2, t3, t2a, acc, tn;
    g2_mul_u64(&t1, p, BLS_U);  g2_neg(&t1, &t1);       /* [z]P */
    g2_psi(&t2, p);                                      /* psi(P) */
    g2_dbl(&t3, p);
    g2_psi(&t3, &t3); g2_psi(&t3, &t3);                  /* psi^2(2P) */
    g2_neg(&tn, &t2);
    g2_add(&t3, &t3, &tn);                               /* psi^2(2P) - psi(P) */
    g2_add(&t2a, &t1, &t2);                              /* zP + psi(P) */
    g2_mul_u64(&t2a, &t2a, BLS_U); g2_neg(&t2a, &t2a);   /* z^2 P + z psi(P) */
    g2_add(&acc, &t3, &t2a);
    g2_neg(&tn, &t1);
    g2_add(&acc, &acc, &tn);                             /* - zP */
    g2_neg(&tn, p);
    g2_add(r, &acc, &tn);                                /* - P */
}

/* setArrayMask + map + cofactor == bls_c_impl.hpp toG for G2 */
static void set_array_mask(fp_t *t, const uint8_t *msg, size_t len) {
    uint8_t buf[48];
    memset(buf, 0, 48);
    size_t n = len < 48 ? len : 48;
    memcpy(buf, msg, n);
    buf[47] &= 0x0F;  /* mask to 380 bits */
    uint64_t raw[6];
    for (int i = 0; i < 6; i++) {
        raw[i] = 0;
        for (int j = 0; j < 8; j++)
            raw[i] |= (uint64_t)buf[i * 8 + j] << (8 * j);
    }
    fp_to_mont(t, raw);
}
int oracle_hash_to_g2_point(g2_t *r, const uint8_t *msg, size_t len) {
    fp2_t t;
    set_array_mask(&t.a, msg, len);
    t.b = FP_ZERO;
    g2aff_t m;
    if (!ft_map_g2(&m, &t)) return 0;
    g2_t mp;
    g2_from_affine(&mp, &m);
    g2_clear_cofactor(r, &mp);
    return 1;
}

/* ================================================================== Fp6 / Fp12 / pairing */
typedef struct { fp2_t c0, c1, c2; } fp6_t;
typedef struct { fp6_t c0, c1; } fp12_t;

static void fp6_add(fp6_t *r, const fp6_t *x, const fp6_t *y) { fp2_add(&r->c0, &x->c0, &y->c0); fp2_add(&r->c1, &x->c1, &y->c1); fp2_add(&r->c2, &x->c2, &y->c2); }
static void fp6_sub(fp6_t *r, const fp6_t *x, const fp6_t *y) { fp2_sub(&r->c0, &x->c0, &y->c0); fp2_sub(&r->c1, &x->c1, &y->c1); fp2_sub(&r->c2, &x->c2, &y->c2); }
static void fp6_neg(fp6_t *r, const fp6_t *x) { fp2_neg(&r->c0, &x->c0); fp2_neg(&r->c1, &x->c1); fp2_neg(&r->c2, &x->c2); }
static void fp6_mul(fp6_t *r, const fp6_t *x, const fp6_t *y) {
    fp2_t t0, t1, t2, s0, s1, tt;
    fp2_mul(&t0, &x->c0, &y->c0);
    fp2_mul(&t1, &x->c1, &y->c1);
    fp2_mul(&t2, &x->c2, &y->c2);
    /* c0 = t0 + xi*((a1+a2)(b1+b2)-t1-t2) */
    fp2_add(&s0, &x->c1, &x->c2);
    fp2_add(&s1, &y->c1, &y->c2);
    fp2_mul(&tt, &s0, &s1);
    fp2_sub(&tt, &tt, &t1);
    fp2_sub(&tt, &tt, &t2);
    fp2_mul_xi(&tt, &tt);
    fp2_t r0; fp2_add(&r0, &t0, &tt);
    /* c1 = (a0+a1)(b0+b1)-t0-t1 + xi*t2 */
    fp2_add(&s0, &x->c0, &x->c1);
    fp2_add(&s1, &y->c0, &y->c1);
    fp2_mul(&tt, &s0, &s1);
    fp2_sub(&tt, &tt, &t0);
    fp2_sub(&tt, &tt, &t1);
    fp2_t xt2; fp2_mul_xi(&xt2, &t2);
    fp2_t r1; fp2_add(&r1, &tt, &xt2);
    /* c2 = (a0+a2)(b0+b2)-t0-t2+t1 */
    fp2_add(&s0, &x->c0, &x->c2);
    fp2_add(&s1, &y->c0, &y->c2);
    fp2_mul(&tt, &s0, &s1);
    fp2_sub(&tt, &tt, &t0);
    fp2_sub(&tt, &tt, &t2);
    fp2_add(&r->c2, &tt, &t1);
    r->c0 = r0; r->c1 = r1;
}
static void fp6_mul_v(fp6_t *r, const fp6_t *x) {
    fp2_t t;
    fp2_mul_xi(&t, &x->c2);
    r->c2 = x->c1; r->c1 = x->c0; r->c0 = t;
}
static void fp6_inv(fp6_t *r, const fp6_t *x) {
    fp2_t c0, c1, c2, t;
    fp2_sqr(&c0, &x->c0);
    fp2_mul(&t, &x->c1, &x->c2);
    fp2_mul_xi(&t, &t);
    fp2_sub(&c0, &c0, &t);
    fp2_sqr(&c1, &x->c2);
    fp2_mul_xi(&c1, &c1);
    fp2_mul(&t, &x->c0, &x->c1);
    fp2_sub(&c1, &c1, &t);
    fp2_sqr(&c2, &x->c1);
    fp2_mul(&t, &x->c0, &x->c2);
    fp2_sub(&c2, &c2, &t);
    /* norm = a0*c0 + xi*(a2*c1 + a1*c2) */
    fp2_t norm, tmp;
    fp2_mul(&norm, &x->c0, &c0);
    fp2_mul(&tmp, &x->c2, &c1);
    fp2_t tmp2; fp2_mul(&tmp2, &x->c1, &c2);
    fp2_add(&tmp, &tmp, &tmp2);
    fp2_mul_xi(&tmp, &tmp);
    fp2_add(&norm, &norm, &tmp);
    fp2_inv(&norm, &norm);
    fp2_mul(&r->c0, &c0, &norm);
    fp2_mul(&r->c1, &c1, &norm);
    fp2_mul(&r->c2, &c2, &norm);
}
static void fp12_mul(fp12_t *r, const fp12_t *x, const fp12_t *y) {
    fp6_t t0, t1, s0, s1, tt;
    fp6_mul(&t0, &x->c0, &y->c0);
    fp6_mul(&t1, &x->c1, &y->c1);
    fp6_add(&s0, &x->c0, &x->c1);
    fp6_add(&s1, &y->c0, &y->c1);
    fp6_mul(&tt, &s0, &s1);
    fp6_sub(&tt, &tt, &t0);
    fp6_sub(&tt, &tt, &t1);
    fp6_t vt1; fp6_mul_v(&vt1, &t1);
    fp6_add(&r->c0, &t0, &vt1);
    r->c1 = tt;
}
static void fp12_sqr(fp12_t *r, const fp12_t *x) {
    /* (a + b w)^2 = (a^2 + v b^2) + 2ab w, via
       a^2 + v b^2 = (a + b)(a + v b) - ab - v ab : 2 fp6 muls total */
    fp6_t ab, apb, avb, t0, vab;
    fp6_mul(&ab, &x->c0, &x->c1);
    fp6_add(&apb, &x->c0, &x->c1);
    fp6_mul_v(&avb, &x->c1);
    fp6_add(&avb, &x->c0, &avb);
    fp6_mul(&t0, &apb, &avb);
    fp6_sub(&t0, &t0, &ab);
    fp6_mul_v(&vab, &ab);
    fp6_sub(&r->c0, &t0, &vab);
    fp6_add(&r->c1, &ab, &ab);
}
static void fp12_conj(fp12_t *r, const fp12_t *x) { r->c0 = x->c0; fp6_neg(&r->c1, &x->c1); }
static void fp12_inv(fp12_t *r, const fp12_t *x) {
    fp6_t t, t1;
    fp6_mul(&t, &x->c0, &x->c0);
    fp6_mul(&t1, &x->c1, &x->c1);
    fp6_mul_v(&t1, &t1);
    fp6_sub(&t, &t, &t1);
    fp6_inv(&t, &t);
    fp6_mul(&r->c0, &x->c0, &t);
    fp6_mul(&t1, &x->c1, &t);
    fp6_neg(&r->c1, &t1);
}
static void fp12_one(fp12_t *r) {
    memset(r, 0, sizeof(*r));
    memcpy(r->c0.c0.a.l, BLS_ONE_P, 48);
}
static int fp12_is_one(const fp12_t *x) {
    fp12_t one;
    fp12_one(&one);
    return memcmp(x, &one, sizeof(one)) == 0;
}
static int fp12_eq(const fp12_t *x, const fp12_t *y) { return memcmp(x, y, sizeof(*x)) == 0; }

/* Frobenius: coeff a_i of w^i -> conj(a_i) * gamma1[i]; layout a = [c0.c0, c1.c0, c0.c1, c1.c1, c0.c2, c1.c2] (w^0..w^5) */
static void fp12_frob(fp12_t *r, const fp12_t *x) {
    const fp2_t *in[6] = { &x->c0.c0, &x->c1.c0, &x->c0.c1, &x->c1.c1, &x->c0.c2, &x->c1.c2 };
    static const uint64_t *GA[6] = { BLS_FROB1_0_A, BLS_FROB1_1_A, BLS_FROB1_2_A, BLS_FROB1_3_A, BLS_FROB1_4_A, BLS_FROB1_5_A };
    static const uint64_t *GB[6] = { BLS_FROB1_0_B, BLS_FROB1_1_B, BLS_FROB1_2_B, BLS_FROB1_3_B, BLS_FROB1_4_B, BLS_FROB1_5_B };
    fp12_t tmp;
    fp2_t *tout[6] = { &tmp.c0.c0, &tmp.c1.c0, &tmp.c0.c1, &tmp.c1.c1, &tmp.c0.c2, &tmp.c1.c2 };
    for (int i = 0; i < 6; i++) {
        fp2_t g, c;
        memcpy(g.a.l, GA[i], 48); memcpy(g.b.l, GB[i], 48);
        fp2_conj(&c, in[i]);
        fp2_mul(tout[i], &c, &g);
    }
    *r = tmp;
}
static void fp12_frob2(fp12_t *r, const fp12_t *x) {
    const fp2_t *in[6] = { &x->c0.c0, &x->c1.c0, &x->c0.c1, &x->c1.c1, &x->c0.c2, &x->c1.c2 };
    static const uint64_t *G[6] = { BLS_FROB2_0, BLS_FROB2_1, BLS_FROB2_2, BLS_FROB2_3, BLS_FROB2_4, BLS_FROB2_5 };
    fp12_t tmp;
    fp2_t *tout[6] = { &tmp.c0.c0, &tmp.c1.c0, &tmp.c0.c1, &tmp.c1.c1, &tmp.c0.c2, &tmp.c1.c2 };
    for (int i = 0; i < 6; i++) {
        fp_t g;
        memcpy(g.l, G[i], 48);
        fp2_mul_fp(tout[i], in[i], &g);
    }
    *r = tmp;
}

/* Miller loop: f_{|z|,Q}(P), Q in E'(Fp2) affine, P in E(Fp) affine.
 * Lines from the Jacobian formulas; l = c0 + c3 w^3 + c5 w^5 where
 *   DBL: c0 = xi*yP*2YZ^3, c3 = 3X^3-2Y^2, c5 = -3X^2 Z^2 xP
 *   ADD: c0 = xi*yP*ZH,    c3 = r*x2 - y2*Z*H, c5 = -r*xP
 * (derivation in DESIGN.md; validated against pyref's embedded-point loop) */
static void fp12_mul_line(fp12_t *f, const fp2_t *c0, const fp2_t *c3, const fp2_t *c5) {
    /* sparse multiply by l = c0 + c3 w^3 + c5 w^5 (l0 = (c0,0,0), l1 = (0,c3,c5)) */
    fp6_t t0, t1, l01, tt, vt1;
    /* t0 = f0 * l0 : scale by c0 */
    fp2_mul(&t0.c0, &f->c0.c0, c0);
    fp2_mul(&t0.c1, &f->c0.c1, c0);
    fp2_mul(&t0.c2, &f->c0.c2, c0);
    /* t1 = f1 * l1 with l1 = (0, c3, c5):
       r0 = xi*(a1*c5 + a2*c3); r1 = xi*a2*c5; r2 = a0*c3 ... careful:
       fp6 mul (a0,a1,a2)*(0,b1,b2) with v^3=xi:
         r0 = xi*(a1*b2 + a2*b1)
         r1 = a0*b1 + xi*(a2*b2)
         r2 = a0*b2 + a1*b1 */
    {
        const fp2_t *a0 = &f->c1.c0, *a1 = &f->c1.c1, *a2 = &f->c1.c2;
        fp2_t p1, p2, q;
        fp2_mul(&p1, a1, c5);
        fp2_mul(&p2, a2, c3);
        fp2_add(&q, &p1, &p2);
        fp2_mul_xi(&t1.c0, &q);
        fp2_mul(&p1, a0, c3);
        fp2_mul(&p2, a2, c5);
        fp2_mul_xi(&p2, &p2);
        fp2_add(&t1.c1, &p1, &p2);
        fp2_mul(&p1, a0, c5);
        fp2_mul(&p2, a1, c3);
        fp2_add(&t1.c2, &p1, &p2);
    }
    /* r1 = (f0+f1)*(l0+l1) - t0 - t1 ; l0+l1 = (c0, c3, c5) */
    fp6_t fs, l;
    fp6_add(&fs, &f->c0, &f->c1);
    l.c0 = *c0; l.c1 = *c3; l.c2 = *c5;
    fp6_mul(&l01, &fs, &l);
    fp6_sub(&tt, &l01, &t0);
    fp6_sub(&tt, &tt, &t1);
    fp6_mul_v(&vt1, &t1);
    fp6_add(&f->c0, &t0, &vt1);
    f->c1 = tt;
}
static void miller_loop(fp12_t *f, const g2aff_t *Q, const g1aff_t *Pa) {
    fp12_one(f);
    g2_t T;
    g2_from_affine(&T, Q);
    /* yP, xP as Fp */
    for (int bit = 62; bit >= 0; bit--) {  /* BLS_U top bit is 63? u = 0xd2..: bit63=1 */
        /* square f */
        fp12_sqr(f, f);
        /* doubling line at T */
        fp2_t A, B, ZZ, c0, c3, c5, t, t2;
        fp2_sqr(&A, &T.x);
        fp2_sqr(&B, &T.y);
        fp2_sqr(&ZZ, &T.z);
        /* c3 = 3X^3 - 2Y^2 = 3A*X - 2B */
        fp2_mul(&t, &A, &T.x);
        fp2_dbl(&t2, &t); fp2_add(&t, &t, &t2);   /* 3X^3 */
        fp2_dbl(&t2, &B);
        fp2_sub(&c3, &t, &t2);
        /* c5 = -3A*ZZ*xP */
        fp2_dbl(&t, &A); fp2_add(&t, &t, &A);     /* 3A */
        fp2_mul(&t, &t, &ZZ);
        fp2_mul_fp(&t, &t, &Pa->x);
        fp2_neg(&c5, &t);
        /* new Z = 2YZ (compute before clobbering T) */
        fp2_t newz;
        fp2_mul(&newz, &T.y, &T.z);
        fp2_dbl(&newz, &newz);
        /* c0 = xi * yP * (2YZ * ZZ) = xi*yP*2YZ^3 */
        fp2_mul(&t, &newz, &ZZ);
        fp2_mul_fp(&t, &t, &Pa->y);
        fp2_mul_xi(&c0, &t);
        /* point doubling (recompute std form) */
        g2_dbl(&T, &T);
        fp12_mul_line(f, &c0, &c3, &c5);
        if ((BLS_U >> bit) & 1) {
            /* addition line: T + Q */
            fp2_t zz, u2, s2, h, rr;
            fp2_sqr(&zz, &T.z);
            fp2_mul(&u2, &Q->x, &zz);
            fp2_mul(&s2, &Q->y, &zz);
            fp2_mul(&s2, &s2, &T.z);
            fp2_sub(&h, &u2, &T.x);
            fp2_sub(&rr, &s2, &T.y);
            /* c0 = xi*yP*(Z*H) */
            fp2_t zh;
            fp2_mul(&zh, &T.z, &h);
            fp2_mul_fp(&t, &zh, &Pa->y);
            fp2_mul_xi(&c0, &t);
            /* c3 = r*x2 - y2*Z*H */
            fp2_mul(&t, &rr, &Q->x);
            fp2_mul(&t2, &Q->y, &zh);
            fp2_sub(&c3, &t, &t2);
            /* c5 = -r*xP */
            fp2_mul_fp(&t, &rr, &Pa->x);
            fp2_neg(&c5, &t);
            /* point: mixed add */
            fp2_t hh, hhh, v;
            fp2_sqr(&hh, &h);
            fp2_mul(&hhh, &hh, &h);
            fp2_mul(&v, &T.x, &hh);
            fp2_sqr(&t, &rr);
            fp2_sub(&t, &t, &hhh);
            fp2_sub(&t, &t, &v);
            fp2_sub(&t, &t, &v);
            fp2_t newx = t;
            fp2_sub(&t, &v, &newx);
            fp2_mul(&t, &rr, &t);
            fp2_mul(&t2, &T.y, &hhh);
            fp2_sub(&T.y, &t, &t2);
            T.x = newx;
            fp2_mul(&T.z, &T.z, &h);
            fp12_mul_line(f, &c0, &c3, &c5);
        }
    }
}

/* Granger-Scott cyclotomic squaring (valid for elements of the cyclotomic
 * subgroup, i.e. after the easy part of the final exponentiation).
 * Coefficients by w-power: a_i at [c0.c0, c1.c0, c0.c1, c1.c1, c0.c2, c1.c2].
 * Working on the Fp4 towers (a0,a3), (a1,a4), (a2,a5):
 *   fp4_sqr((a,b)) = (a^2 + xi*b^2... ) per GS2010. */
static void fp4_sqr(fp2_t *c, fp2_t *d, const fp2_t *a, const fp2_t *b) {
    /* (c + d*t) = (a + b*t)^2 in Fp4 = Fp2[t]/(t^2 - xi):
       c = a^2 + xi*b^2, d = 2ab */
    fp2_t a2, b2, t;
    fp2_sqr(&a2, a);
    fp2_sqr(&b2, b);
    fp2_mul_xi(&t, &b2);
    fp2_add(c, &a2, &t);
    fp2_add(&t, a, b);
    fp2_sqr(&t, &t);
    fp2_sub(&t, &t, &a2);
    fp2_sub(d, &t, &b2);
}
static void fp12_cyc_sqr(fp12_t *r, const fp12_t *x) {
    const fp2_t *a0 = &x->c0.c0, *a1 = &x->c1.c0, *a2 = &x->c0.c1,
                *a3 = &x->c1.c1, *a4 = &x->c0.c2, *a5 = &x->c1.c2;
    fp2_t t00, t03, t01, t04, t02, t05;
    fp4_sqr(&t00, &t03, a0, a3);   /* (a0 + a3 t) */
    fp4_sqr(&t01, &t04, a1, a4);   /* (a1 + a4 t) */
    fp4_sqr(&t02, &t05, a2, a5);   /* (a2 + a5 t) */
    /* r0 = 3 t00 - 2 a0 ; r3 = 3 t03 + 2 a3
       r2 = 3 t01 - 2 a2 ; r5 = 3 t04 + 2 a5   (shifted by the w-multiplication)
       r4 = 3 t02 - 2 a4 ; r1 = 3 xi*t05 + 2 a1 */
    fp12_t out;
    fp2_t tmp;
    /* r0 = 2(t00 - a0) + t00 */
    fp2_sub(&tmp, &t00, a0); fp2_dbl(&tmp, &tmp); fp2_add(&out.c0.c0, &tmp, &t00);
    /* r2 (coeff a2, at w^2) = 2(t01 - a2) + t01 */
    fp2_sub(&tmp, &t01, a2); fp2_dbl(&tmp, &tmp); fp2_add(&out.c0.c1, &tmp, &t01);
    /* r4 (a4, w^4) = 2(t02 - a4) + t02 */
    fp2_sub(&tmp, &t02, a4); fp2_dbl(&tmp, &tmp); fp2_add(&out.c0.c2, &tmp, &t02);
    /* r1 (a1, w^1) = 2(xi*t05 + a1) + xi*t05 */
    fp2_t x05;
    fp2_mul_xi(&x05, &t05);
    fp2_add(&tmp, &x05, a1); fp2_dbl(&tmp, &tmp); fp2_add(&out.c1.c0, &tmp, &x05);
    /* r3 (a3, w^3) = 2(t03 + a3) + t03 */
    fp2_add(&tmp, &t03, a3); fp2_dbl(&tmp, &tmp); fp2_add(&out.c1.c1, &tmp, &t03);
    /* r5 (a5, w^5) = 2(t04 + a5) + t04 */
    fp2_add(&tmp, &t04, a5); fp2_dbl(&tmp, &tmp); fp2_add(&out.c1.c2, &tmp, &t04);
    *r = out;
}
/* exp by |z|; cyclotomic squarings (callers apply this only after the easy
 * part, where x is in the cyclotomic subgroup). */
static void fp12_pow_u(fp12_t *r, const fp12_t *x) {
    fp12_t acc = *x;
    for (int bit = 62; bit >= 0; bit--) {
        fp12_cyc_sqr(&acc, &acc);
        if ((BLS_U >> bit) & 1) fp12_mul(&acc, &acc, x);
    }
    *r = acc;
}

/* final exponentiation: easy part, then HHT hard part:
 * 3(p^4-p^2+1)/r = (z-1)^2 (z+p) (z^2+p^2-1) + 3  (identity asserted in
 * gen_constants.py).  Output is e(P,Q)^3 — boolean ==1 semantics unchanged
 * (gcd(3, r) = 1); oracle and HIP path use the identical convention. */
static void final_exp(fp12_t *r, const fp12_t *f_in) {
    fp12_t f, t, inv;
    /* easy: f = f^(p^6-1) = conj(f) * f^-1; then f = f^(p^2) * f */
    fp12_conj(&t, f_in);
    fp12_inv(&inv, f_in);
    fp12_mul(&f, &t, &inv);
    fp12_frob2(&t, &f);
    fp12_mul(&f, &t, &f);
    /* hard: a = f^(z-1) = conj(f^u * f); b = a^(z-1); c = b^z * b^p;
     * d = c^(z^2) * frob2(c) * conj(c); out = d * f^3 */
    fp12_t a, b, c, d, u1, u2;
    fp12_pow_u(&u1, &f);
    fp12_mul(&a, &u1, &f);
    fp12_conj(&a, &a);
    fp12_pow_u(&u1, &a);
    fp12_mul(&b, &u1, &a);
    fp12_conj(&b, &b);
    fp12_pow_u(&u1, &b);
    fp12_conj(&u1, &u1);          /* b^z */
    fp12_frob(&u2, &b);           /* b^p */
    fp12_mul(&c, &u1, &u2);
    fp12_pow_u(&u1, &c);
    fp12_pow_u(&u1, &u1);         /* c^(z^2) = c^(u^2), sign cancels */
    fp12_frob2(&u2, &c);
    fp12_mul(&d, &u1, &u2);
    fp12_conj(&u1, &c);           /* c^-1 (cyclotomic) */
    fp12_mul(&d, &d, &u1);
    fp12_sqr(&t, &f);
    fp12_mul(&t, &t, &f);         /* f^3 */
    fp12_mul(r, &d, &t);
}

/* pairing product check: finalExp( conj(prod_i ML(Q_i, P_i)) ) == 1.
 * z<0: conjugation applied once after the product (conj is a homomorphism). */
int oracle_pairing_check2(const g2aff_t *Q1, const g1aff_t *P1,
                          const g2aff_t *Q2, const g1aff_t *P2) {
    /* check e(P1,Q1) * e(P2,Q2) == 1 (caller negates one side) */
    fp12_t f1, f2, f;
    miller_loop(&f1, Q1, P1);
    miller_loop(&f2, Q2, P2);
    fp12_mul(&f, &f1, &f2);
    fp12_conj(&f, &f);
    final_exp(&f, &f);
    return fp12_is_one(&f);
}

/* test support: both G2 membership methods on a compressed candidate */
int oracle_g2_subgroup_methods(const uint8_t in96[96], int32_t out2[2]) {
    g2_t p;
    if (!oracle_g2_deserialize(&p, in96, 0)) { out2[0] = out2[1] = -1; return 0; }
    out2[0] = g2_in_subgroup(&p);
    out2[1] = g2_in_subgroup_fast(&p);
    return 1;
}

/* self-test: cyclotomic squaring must equal the full squaring on elements of
 * the cyclotomic subgroup (exercised by tests/test_oracle.py) */
int oracle_test_cyc_sqr(void) {
    g1_t base;
    g1aff_t ba;
    g2aff_t qa;
    fp_base_point(&base);
    g1_to_affine(&ba, &base);
    memcpy(qa.x.a.l, BLS_G2_XA, 48); memcpy(qa.x.b.l, BLS_G2_XB, 48);
    memcpy(qa.y.a.l, BLS_G2_YA, 48); memcpy(qa.y.b.l, BLS_G2_YB, 48);
    fp12_t f, easy, t, inv, a, b;
    miller_loop(&f, &qa, &ba);
    fp12_conj(&t, &f);
    fp12_inv(&inv, &f);
    fp12_mul(&easy, &t, &inv);
    fp12_frob2(&t, &easy);
    fp12_mul(&easy, &t, &easy);
    fp12_cyc_sqr(&a, &easy);
    fp12_sqr(&b, &easy);
    return fp12_eq(&a, &b);
}

/* ================================================================== Keccak-256 */
static const uint64_t KECCAK_RC[24] = {
    0x0000000000000001ULL, 0x0000000000008082ULL, 0x800000000000808aULL, 0x8000000080008000ULL,
    0x000000000000808bULL, 0x0000000080000001ULL, 0x8000000080008081ULL, 0x8000000000008009ULL,
    0x000000000000008aULL, 0x0000000000000088ULL, 0x0000000080008009ULL, 0x000000008000000aULL,
    0x000000008000808bULL, 0x800000000000008bULL, 0x8000000000008089ULL, 0x8000000000008003ULL,
    0x8000000000008002ULL, 0x8000000000000080ULL, 0x000000000000800aULL, 0x800000008000000aULL,
    0x8000000080008081ULL, 0x8000000000008080ULL, 0x0000000080000001ULL, 0x8000000080008008ULL };
static inline uint64_t rol64(uint64_t v, int s) { return s ? (v << s) | (v >> (64 - s)) : v; }
static void keccak_f(uint64_t st[25]) {
    static const int rot[25] = { 0,1,62,28,27, 36,44,6,55,20, 3,10,43,25,39, 41,45,15,21,8, 18,2,61,56,14 };
    static const int pi[25] = { 0,6,12,18,24, 3,9,10,16,22, 1,7,13,19,20, 4,5,11,17,23, 2,8,14,15,21 };
    for (int rnd = 0; rnd < 24; rnd++) {
        uint64_t C[5], D[5], B[25];
        for (int x = 0; x < 5; x++)
            C[x] = st[x] ^ st[x + 5] ^ st[x + 10] ^ st[x + 15] ^ st[x + 20];
        for (int x = 0; x < 5; x++)
            D[x] = C[(x + 4) % 5] ^ rol64(C[(x + 1) % 5], 1);
        for (int i = 0; i < 25; i++) st[i] ^= D[i % 5];
        for (int i = 0; i < 25; i++) B[i] = rol64(st[pi[i]], rot[pi[i]]);
        for (int y = 0; y < 5; y++)
            for (int x = 0; x < 5; x++)
                st[y * 5 + x] = B[y * 5 + x] ^ ((~B[y * 5 + (x + 1) % 5]) & B[y * 5 + (x + 2) % 5]);
        st[0] ^= KECCAK_RC[rnd];
    }
}
void oracle_keccak256(const uint8_t *in, size_t len, uint8_t out[32]) {
    uint64_t st[25];
    memset(st, 0, sizeof(st));
    const size_t rate = 136;
    uint8_t blk[136];
    while (len >= rate) {
        for (size_t i = 0; i < rate / 8; i++) {
            uint64_t v = 0;
            for (int j = 0; j < 8; j++) v |= (uint64_t)in[8 * i + j] << (8 * j);
            st[i] ^= v;
        }
        keccak_f(st);
        in += rate; len -= rate;
    }
    memset(blk, 0, rate);
    memcpy(blk, in, len);
    blk[len] = 0x01;
    blk[rate - 1] |= 0x80;
    for (size_t i = 0; i < rate / 8; i++) {
        uint64_t v = 0;
        for (int j = 0; j < 8; j++) v |= (uint64_t)blk[8 * i + j] << (8 * j);
        st[i] ^= v;
    }
    keccak_f(st);
    for (int i = 0; i < 4; i++)
        for (int j = 0; j < 8; j++)
            out[8 * i + j] = (uint8_t)(st[i] >> (8 * j));
}

/* ================================================================== public serialized API */
static int fr_from_le32(uint64_t k[4], const uint8_t in[32]) {
    for (int i = 0; i < 4; i++) {
        k[i] = 0;
        for (int j = 0; j < 8; j++) k[i] |= (uint64_t)in[i * 8 + j] << (8 * j);
    }
    for (int i = 3; i >= 0; i--) {
        if (k[i] > BLS_R[i]) return 0;
        if (k[i] < BLS_R[i]) return 1;
    }
    return 0; /* == r rejected */
}

int oracle_pk_from_sk(const uint8_t sk32[32], uint8_t pk48[48]) {
    uint64_t k[4];
    if (!fr_from_le32(k, sk32)) return 0;
    g1_t base, pk;
    fp_base_point(&base);
    g1_mul(&pk, &base, k, 4);
    oracle_g1_serialize(pk48, &pk);
    return 1;
}
int oracle_sign_hash(const uint8_t sk32[32], const uint8_t *msg, size_t len, uint8_t sig96[96]) {
    uint64_t k[4];
    if (!fr_from_le32(k, sk32)) return 0;
    g2_t hm, sig;
    if (!oracle_hash_to_g2_point(&hm, msg, len)) return 0;
    g2_mul(&sig, &hm, k, 4);
    oracle_g2_serialize(sig96, &sig);
    return 1;
}
int oracle_hash_to_g2(const uint8_t *msg, size_t len, uint8_t out96[96]) {
    g2_t hm;
    if (!oracle_hash_to_g2_point(&hm, msg, len)) return 0;
    oracle_g2_serialize(out96, &hm);
    return 1;
}
/* returns 1 accept, 0 reject, -1 deserialization error */
int oracle_verify_hash(const uint8_t pk48[48], const uint8_t sig96[96],
                       const uint8_t *msg, size_t len) {
    g1_t pub;
    g2_t sig, hm;
    if (!oracle_g1_deserialize(&pub, pk48, 1)) return -1;
    if (!oracle_g2_deserialize(&sig, sig96, 1)) return -1;
    if (!oracle_hash_to_g2_point(&hm, msg, len)) return -1;
    /* e(pub, Hm) == e(base, sig)  <=>  e(pub,Hm) * e(-base, sig) == 1 */
    if (g1_is_inf(&pub) && g2_is_inf(&sig)) return 1;   /* herumi edge */
    if (g1_is_inf(&pub) || g2_is_inf(&sig)) return 0;
    g1_t nbase; g1aff_t pa, ba;
    g2aff_t ha, sa;
    fp_base_point(&nbase);
    g1_neg(&nbase, &nbase);
    g1_to_affine(&pa, &pub);
    g1_to_affine(&ba, &nbase);
    g2_to_affine(&ha, &hm);
    g2_to_affine(&sa, &sig);
    return oracle_pairing_check2(&ha, &pa, &sa, &ba);
}
/* PublicKey.Add/Sub on serialized keys; invalid input -> 0 */
int oracle_g1_add_ser(const uint8_t a48[48], const uint8_t b48[48], uint8_t out48[48], int sub) {
    g1_t a, b;
    if (!oracle_g1_deserialize(&a, a48, 1)) return 0;
    if (!oracle_g1_deserialize(&b, b48, 1)) return 0;
    if (sub) g1_neg(&b, &b);
    g1_add(&a, &a, &b);
    oracle_g1_serialize(out48, &a);
    return 1;
}
int oracle_g2_add_ser(const uint8_t a96[96], const uint8_t b96[96], uint8_t out96[96], int sub) {
    g2_t a, b;
    if (!oracle_g2_deserialize(&a, a96, 1)) return 0;
    if (!oracle_g2_deserialize(&b, b96, 1)) return 0;
    if (sub) g2_neg(&b, &b);
    g2_add(&a, &a, &b);
    oracle_g2_serialize(out96, &a);
    return 1;
}
int oracle_g1_deserialize_check(const uint8_t in48[48]) {
    g1_t p;
    return oracle_g1_deserialize(&p, in48, 1);
}
int oracle_g2_deserialize_check(const uint8_t in96[96]) {
    g2_t p;
    return oracle_g2_deserialize(&p, in96, 1);
}
/* Mask.SetMask masked aggregate: sum of pk_i where bit i set (mask.go:113-134).
 * pks48cat = n*48 bytes. */
int oracle_mask_aggregate_pub(const uint8_t *pks48cat, const uint8_t *bitmap,
                              size_t n, uint8_t out48[48]) {
    g1_t acc;
    g1_set_inf(&acc);
    for (size_t i = 0; i < n; i++) {
        if ((bitmap[i >> 3] >> (i & 7)) & 1) {
            g1_t p;
            if (!oracle_g1_deserialize(&p, pks48cat + 48 * i, 1)) return 0;
            g1_add(&acc, &acc, &p);
        }
    }
    oracle_g1_serialize(out48, &acc);
    return 1;
}
int oracle_aggregate_sigs(const uint8_t *sigs96cat, size_t n, uint8_t out96[96]) {
    g2_t acc;
    g2_set_inf(&acc);
    for (size_t i = 0; i < n; i++) {
        g2_t s;
        if (!oracle_g2_deserialize(&s, sigs96cat + 96 * i, 1)) return 0;
        g2_add(&acc, &acc, &s);
    }
    oracle_g2_serialize(out96, &acc);
    return 1;
}
/* one aggregate-verify (the north-star unit): masked pubkey sum over the
 * committee + pairing check of the aggregate signature on msg.
 * pks are PRE-VALIDATED serialized keys (subgroup check done at table build,
 * mirroring the reference's LRU-cached deserialization, mask.go:13-15). */
int oracle_agg_verify(const uint8_t *pks48cat, const uint8_t *bitmap, size_t n,
                      const uint8_t sig96[96], const uint8_t *msg, size_t mlen) {
    g1_t acc;
    g1_set_inf(&acc);
    for (size_t i = 0; i < n; i++) {
        if ((bitmap[i >> 3] >> (i & 7)) & 1) {
            g1_t p;
            if (!oracle_g1_deserialize(&p, pks48cat + 48 * i, 0)) return -1;
            g1_add(&acc, &acc, &p);
        }
    }
    uint8_t agg48[48];
    oracle_g1_serialize(agg48, &acc);
    return oracle_verify_hash(agg48, sig96, msg, mlen);
}
/* ---- committee table: pubkeys decompressed+validated ONCE (mirrors the
 * reference's LRU-cached PublicKeyWrapper.Object, crypto/bls/bls.go:30-33,
 * mask.go:13-15); mask-sum then uses mixed (Jacobian+affine) adds. */
#include <stdlib.h>
typedef struct { g1aff_t *pts; size_t n; g1_t full_sum; } committee_t;

void *oracle_committee_build(const uint8_t *pks48cat, size_t n) {
    committee_t *c = (committee_t *)malloc(sizeof(committee_t));
    c->pts = (g1aff_t *)malloc(sizeof(g1aff_t) * n);
    c->n = n;
    for (size_t i = 0; i < n; i++) {
        g1_t p;
        if (!oracle_g1_deserialize(&p, pks48cat + 48 * i, 1)) {
            free(c->pts); free(c);
            return NULL;
        }
        g1_to_affine(&c->pts[i], &p);   /* keys are never infinity in practice */
    }
    /* committee-wide sum, for the dense-mask complement path (same trick as
     * the GPU mask kernel: group-exact, so serialized results are identical) */
    g1_set_inf(&c->full_sum);
    {
        g1_t acc; g1_set_inf(&acc);
        for (size_t i = 0; i < n; i++) {
            /* forward-declared below */
            extern void oracle_g1_madd_fwd(g1_t *r, const g1_t *p, const g1aff_t *q);
            oracle_g1_madd_fwd(&acc, &acc, &c->pts[i]);
        }
        c->full_sum = acc;
    }
    return c;
}
void oracle_committee_free(void *h) {
    committee_t *c = (committee_t *)h;
    if (c) { free(c->pts); free(c); }
}
/* mixed add: Jacobian += affine */
static void g1_madd(g1_t *r, const g1_t *p, const g1aff_t *q) {
    if (g1_is_inf(p)) { g1_from_affine(r, q); return; }
    fp_t z1z1, u2, s2, h, rr, hh, hhh, v, t;
    fp_sqr(&z1z1, &p->z);
    fp_mul(&u2, &q->x, &z1z1);
    fp_mul(&s2, &q->y, &p->z);
    fp_mul(&s2, &s2, &z1z1);
    fp_sub(&h, &u2, &p->x);
    fp_sub(&rr, &s2, &p->y);
    if (fp_is_zero(&h)) {
        if (fp_is_zero(&rr)) { g1_dbl(r, p); return; }
        g1_set_inf(r); return;
    }
    fp_sqr(&hh, &h);
    fp_mul(&hhh, &hh, &h);
    fp_mul(&v, &p->x, &hh);
    fp_sqr(&t, &rr);
    fp_sub(&t, &t, &hhh);
    fp_sub(&t, &t, &v);
    fp_sub(&r->x, &t, &v);
    fp_sub(&t, &v, &r->x);
    fp_mul(&t, &rr, &t);
    fp_mul(&v, &p->y, &hhh);
    fp_sub(&r->y, &t, &v);
    fp_mul(&t, &p->z, &h);
    r->z = t;
}
void oracle_g1_madd_fwd(g1_t *r, const g1_t *p, const g1aff_t *q) { g1_madd(r, p, q); }

/* one aggregate-verify against a prebuilt table (the north-star unit) */
static void masked_sum_tab(const committee_t *c, const uint8_t *bitmap, g1_t *out) {
    size_t cnt = 0;
    for (size_t i = 0; i < c->n; i++)
        cnt += (bitmap[i >> 3] >> (i & 7)) & 1;
    int complement = cnt > c->n / 2;
    g1_t acc;
    g1_set_inf(&acc);
    for (size_t i = 0; i < c->n; i++) {
        int bit = (bitmap[i >> 3] >> (i & 7)) & 1;
        if (bit != complement)
            g1_madd(&acc, &acc, &c->pts[i]);
    }
    if (complement) {
        g1_neg(&acc, &acc);
        g1_add(out, &c->full_sum, &acc);
    } else {
        *out = acc;
    }
}
int oracle_agg_verify_tab(const void *h, const uint8_t *bitmap,
                          const uint8_t sig96[96], const uint8_t *msg, size_t mlen) {
    const committee_t *c = (const committee_t *)h;
    g1_t acc;
    masked_sum_tab(c, bitmap, &acc);
    uint8_t agg48[48];
    oracle_g1_serialize(agg48, &acc);
    return oracle_verify_hash(agg48, sig96, msg, mlen);
}
int oracle_batch_agg_verify_tab(const void *h, const uint8_t *bitmaps,
                                const uint8_t *sigs96, const uint8_t *msgs,
                                size_t mlen, size_t batch, int32_t *results) {
    const committee_t *c = (const committee_t *)h;
    size_t bm_len = (c->n + 7) / 8;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
    for (size_t j = 0; j < batch; j++)
        results[j] = oracle_agg_verify_tab(h, bitmaps + j * bm_len,
                                           sigs96 + j * 96, msgs + j * mlen, mlen);
    return 1;
}
/* masked aggregate only (Mask.SetMask equivalent) from the table */
int oracle_mask_aggregate_tab(const void *h, const uint8_t *bitmap, uint8_t out48[48]) {
    const committee_t *c = (const committee_t *)h;
    g1_t acc;
    masked_sum_tab(c, bitmap, &acc);
    oracle_g1_serialize(out48, &acc);
    return 1;
}

/* batch of independent aggregate-verifies (CPU baseline unit); OpenMP across items */
int oracle_batch_agg_verify(const uint8_t *pks48cat, size_t n,
                            const uint8_t *bitmaps, const uint8_t *sigs96,
                            const uint8_t *msgs, size_t mlen, size_t batch,
                            int32_t *results) {
    size_t bm_len = (n + 7) / 8;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
    for (size_t j = 0; j < batch; j++) {
        results[j] = oracle_agg_verify(pks48cat, bitmaps + j * bm_len, n,
                                       sigs96 + j * 96, msgs + j * mlen, mlen);
    }
    return 1;
}
int oracle_nthreads(void) {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}
