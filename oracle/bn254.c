/* oracle/bn254.c — CPU oracle: BN254 field/curve arithmetic, Pippenger MSM,
 * radix-2 NTT. TEST INFRASTRUCTURE ONLY (see bn254.h header).
 *
 * Restates, independently (no reference code copied — the reference's
 * arithmetic lives in un-vendored Rust crates and was not even readable as
 * source in this environment):
 *   - `best_multiexp(coeffs, bases)` of halo2curves-axiom 0.5.2 (declared in
 *     /root/reference/Cargo.toml:53; called transitively from
 *     lightclient-circuits/src/util/circuit.rs:158,177,211): computes
 *     Sum_i coeffs_i * bases_i over G1. The value is windowing-independent,
 *     so this Pippenger restatement uses its own window width.
 *   - `best_fft(a, omega, log_n)` of the PSE halo2_proofs fork: in-place
 *     radix-2 DFT, out[j] = Sum_i a[i] omega^(ij); EvaluationDomain::ifft
 *     additionally scales by n^{-1}; coset variants multiply by powers of
 *     the coset generator before/after (see oracle_ntt_fr contract).
 *   - halo2curves 4x64-limb little-endian Montgomery memory format (R=2^256).
 *
 * Implementation: 4x64-bit limbs with unsigned __int128 products (the HIP
 * kernels use 8x32-bit limbs — deliberately a different decomposition so the
 * two implementations cannot share a limb-level bug), CIOS Montgomery
 * multiplication, Jacobian coordinates, OpenMP over Pippenger windows and
 * NTT butterfly blocks.
 */
#include "bn254.h"
#include <stdlib.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

typedef unsigned __int128 u128;
typedef struct { uint64_t l[4]; } fe;
typedef struct { fe mod; uint64_t inv; fe r2; fe one; } fctx;

/* constants computed from the published alt_bn128 moduli */
static const fctx FQ = {
    .mod = {{0x3c208c16d87cfd47ull, 0x97816a916871ca8dull, 0xb85045b68181585dull, 0x30644e72e131a029ull}},
    .inv = 0x87d20782e4866389ull,
    .r2  = {{0xf32cfc5b538afa89ull, 0xb5e71911d44501fbull, 0x47ab1eff0a417ff6ull, 0x06d89f71cab8351full}},
    .one = {{0xd35d438dc58f0d9dull, 0x0a78eb28f5c70b3dull, 0x666ea36f7879462cull, 0x0e0a77c19a07df2full}},
};
static const fctx FR = {
    .mod = {{0x43e1f593f0000001ull, 0x2833e84879b97091ull, 0xb85045b68181585dull, 0x30644e72e131a029ull}},
    .inv = 0xc2e1f593efffffffull,
    .r2  = {{0x1bb8e645ae216da7ull, 0x53fe3ab1e35c59e3ull, 0x8c49833d53bb8085ull, 0x0216d0b17f4e44a5ull}},
    .one = {{0xac96341c4ffffffbull, 0x36fc76959f60cd29ull, 0x666ea36f7879462eull, 0x0e0a77c19a07df2full}},
};

/* ------------------------------------------------------------ core bignum */
static inline int fe_is_zero(const fe* a) {
    return (a->l[0] | a->l[1] | a->l[2] | a->l[3]) == 0;
}
static inline int fe_eq(const fe* a, const fe* b) {
    return ((a->l[0]^b->l[0]) | (a->l[1]^b->l[1]) | (a->l[2]^b->l[2]) | (a->l[3]^b->l[3])) == 0;
}
static inline int fe_geq(const fe* a, const fe* b) { /* a >= b */
    for (int i = 3; i >= 0; i--) {
        if (a->l[i] != b->l[i]) return a->l[i] > b->l[i];
    }
    return 1;
}
static inline uint64_t fe_sub_raw(fe* o, const fe* a, const fe* b) { /* returns borrow */
    u128 brw = 0;
    for (int i = 0; i < 4; i++) {
        u128 d = (u128)a->l[i] - b->l[i] - (uint64_t)brw;
        o->l[i] = (uint64_t)d;
        brw = (d >> 64) & 1; /* two's complement borrow */
    }
    return (uint64_t)brw;
}
static inline uint64_t fe_add_raw(fe* o, const fe* a, const fe* b) { /* returns carry */
    u128 c = 0;
    for (int i = 0; i < 4; i++) {
        c += (u128)a->l[i] + b->l[i];
        o->l[i] = (uint64_t)c;
        c >>= 64;
    }
    return (uint64_t)c;
}
static void f_add(const fctx* f, fe* o, const fe* a, const fe* b) {
    uint64_t c = fe_add_raw(o, a, b);
    if (c || fe_geq(o, &f->mod)) fe_sub_raw(o, o, &f->mod);
}
static void f_sub(const fctx* f, fe* o, const fe* a, const fe* b) {
    if (fe_sub_raw(o, a, b)) fe_add_raw(o, o, &f->mod);
}
static void f_neg(const fctx* f, fe* o, const fe* a) {
    if (fe_is_zero(a)) { *o = *a; return; }
    fe_sub_raw(o, &f->mod, a);
}
/* CIOS Montgomery multiplication, 4x64 */
static void f_mul(const fctx* f, fe* o, const fe* a, const fe* b) {
    uint64_t t[6] = {0, 0, 0, 0, 0, 0};
    for (int i = 0; i < 4; i++) {
        u128 c = 0;
        for (int j = 0; j < 4; j++) {
            c += (u128)a->l[i] * b->l[j] + t[j];
            t[j] = (uint64_t)c;
            c >>= 64;
        }
        c += t[4];
        t[4] = (uint64_t)c;
        t[5] = (uint64_t)(c >> 64);
        uint64_t m = t[0] * f->inv;
        c = (u128)m * f->mod.l[0] + t[0];
        c >>= 64;
        for (int j = 1; j < 4; j++) {
            c += (u128)m * f->mod.l[j] + t[j];
            t[j - 1] = (uint64_t)c;
            c >>= 64;
        }
        c += t[4];
        t[3] = (uint64_t)c;
        t[4] = t[5] + (uint64_t)(c >> 64);
    }
    fe r = {{t[0], t[1], t[2], t[3]}};
    if (t[4] || fe_geq(&r, &f->mod)) fe_sub_raw(&r, &r, &f->mod);
    *o = r;
}
static void f_sqr(const fctx* f, fe* o, const fe* a) { f_mul(f, o, a, a); }
/* to/from Montgomery */
static void f_to_mont(const fctx* f, fe* o, const fe* a)   { f_mul(f, o, a, &f->r2); }
static void f_from_mont(const fctx* f, fe* o, const fe* a) { fe one = {{1,0,0,0}}; f_mul(f, o, a, &one); }
/* exponentiation by canonical 256-bit exponent (Montgomery base/out) */
static void f_pow(const fctx* f, fe* o, const fe* a, const fe* e) {
    fe acc = f->one, base = *a;
    for (int i = 0; i < 256; i++) {
        if ((e->l[i >> 6] >> (i & 63)) & 1) f_mul(f, &acc, &acc, &base);
        f_sqr(f, &base, &base);
    }
    *o = acc;
}
static void f_inv(const fctx* f, fe* o, const fe* a) { /* Fermat: a^(m-2) */
    fe e = f->mod;
    /* m - 2: both moduli are odd and > 2, low limb can't underflow twice */
    e.l[0] -= 2;
    f_pow(f, o, a, &e);
}
static void fe_from_bytes(fe* o, const uint8_t* b) { memcpy(o->l, b, 32); }
static void fe_to_bytes(uint8_t* b, const fe* a)   { memcpy(b, a->l, 32); }

/* ------------------------------------------------------------ G1 Jacobian */
typedef struct { fe x, y; int inf; } g1a;
typedef struct { fe X, Y, Z; } g1j;            /* Z==0 => identity */

static const fe FE_ZERO = {{0, 0, 0, 0}};

static void g1a_from_bytes(g1a* p, const uint8_t* b) {
    fe_from_bytes(&p->x, b);
    fe_from_bytes(&p->y, b + 32);
    p->inf = fe_is_zero(&p->x) && fe_is_zero(&p->y);
}
static void g1a_to_bytes(uint8_t* b, const g1a* p) {
    if (p->inf) { memset(b, 0, 64); return; }
    fe_to_bytes(b, &p->x);
    fe_to_bytes(b + 32, &p->y);
}
static void g1j_set_inf(g1j* p) { p->X = FQ.one; p->Y = FQ.one; p->Z = FE_ZERO; }
static int  g1j_is_inf(const g1j* p) { return fe_is_zero(&p->Z); }
static void g1j_from_affine(g1j* o, const g1a* p) {
    if (p->inf) { g1j_set_inf(o); return; }
    o->X = p->x; o->Y = p->y; o->Z = FQ.one;
}
static void g1j_to_affine(g1a* o, const g1j* p) {
    if (g1j_is_inf(p)) { o->x = FE_ZERO; o->y = FE_ZERO; o->inf = 1; return; }
    fe zi, zi2, zi3;
    f_inv(&FQ, &zi, &p->Z);
    f_sqr(&FQ, &zi2, &zi);
    f_mul(&FQ, &zi3, &zi2, &zi);
    f_mul(&FQ, &o->x, &p->X, &zi2);
    f_mul(&FQ, &o->y, &p->Y, &zi3);
    o->inf = 0;
}
/* dbl-2009-l (a=0) */
static void g1j_dbl(g1j* o, const g1j* p) {
    if (g1j_is_inf(p)) { *o = *p; return; }
    fe A, B, C, D, E, F, t;
    f_sqr(&FQ, &A, &p->X);
    f_sqr(&FQ, &B, &p->Y);
    f_sqr(&FQ, &C, &B);
    f_add(&FQ, &D, &p->X, &B);
    f_sqr(&FQ, &D, &D);
    f_sub(&FQ, &D, &D, &A);
    f_sub(&FQ, &D, &D, &C);
    f_add(&FQ, &D, &D, &D);
    f_add(&FQ, &E, &A, &A);
    f_add(&FQ, &E, &E, &A);
    f_sqr(&FQ, &F, &E);
    f_mul(&FQ, &t, &p->Y, &p->Z);
    f_add(&FQ, &o->Z, &t, &t);
    f_sub(&FQ, &o->X, &F, &D);
    f_sub(&FQ, &o->X, &o->X, &D);
    f_sub(&FQ, &t, &D, &o->X);
    f_mul(&FQ, &t, &E, &t);
    f_add(&FQ, &C, &C, &C);
    f_add(&FQ, &C, &C, &C);
    f_add(&FQ, &C, &C, &C);
    f_sub(&FQ, &o->Y, &t, &C);
}
/* mixed add, madd-2007-bl, with degenerate-case handling */
static void g1j_add_affine(g1j* o, const g1j* p, const g1a* q) {
    if (q->inf) { *o = *p; return; }
    if (g1j_is_inf(p)) { g1j_from_affine(o, q); return; }
    fe Z1Z1, U2, S2, H, HH, I, J, rr, V, t;
    f_sqr(&FQ, &Z1Z1, &p->Z);
    f_mul(&FQ, &U2, &q->x, &Z1Z1);
    f_mul(&FQ, &S2, &q->y, &p->Z);
    f_mul(&FQ, &S2, &S2, &Z1Z1);
    f_sub(&FQ, &H, &U2, &p->X);
    f_sub(&FQ, &rr, &S2, &p->Y);
    if (fe_is_zero(&H)) {
        if (fe_is_zero(&rr)) { g1j_dbl(o, p); return; }
        g1j_set_inf(o);
        return;
    }
    f_add(&FQ, &rr, &rr, &rr);
    f_sqr(&FQ, &HH, &H);
    f_add(&FQ, &I, &HH, &HH);
    f_add(&FQ, &I, &I, &I);
    f_mul(&FQ, &J, &H, &I);
    f_mul(&FQ, &V, &p->X, &I);
    f_sqr(&FQ, &o->X, &rr);
    f_sub(&FQ, &o->X, &o->X, &J);
    f_sub(&FQ, &o->X, &o->X, &V);
    f_sub(&FQ, &o->X, &o->X, &V);
    f_sub(&FQ, &t, &V, &o->X);
    f_mul(&FQ, &t, &rr, &t);
    f_mul(&FQ, &J, &p->Y, &J);
    f_add(&FQ, &J, &J, &J);
    fe Ynew;
    f_sub(&FQ, &Ynew, &t, &J);
    f_add(&FQ, &t, &p->Z, &H);
    f_sqr(&FQ, &t, &t);
    f_sub(&FQ, &t, &t, &Z1Z1);
    f_sub(&FQ, &o->Z, &t, &HH);
    o->Y = Ynew;
}
/* general Jacobian add, add-2007-bl, with degenerate-case handling */
static void g1j_add(g1j* o, const g1j* p, const g1j* q) {
    if (g1j_is_inf(p)) { *o = *q; return; }
    if (g1j_is_inf(q)) { *o = *p; return; }
    fe Z1Z1, Z2Z2, U1, U2, S1, S2, H, I, J, rr, V, t;
    f_sqr(&FQ, &Z1Z1, &p->Z);
    f_sqr(&FQ, &Z2Z2, &q->Z);
    f_mul(&FQ, &U1, &p->X, &Z2Z2);
    f_mul(&FQ, &U2, &q->X, &Z1Z1);
    f_mul(&FQ, &S1, &p->Y, &q->Z);
    f_mul(&FQ, &S1, &S1, &Z2Z2);
    f_mul(&FQ, &S2, &q->Y, &p->Z);
    f_mul(&FQ, &S2, &S2, &Z1Z1);
    f_sub(&FQ, &H, &U2, &U1);
    f_sub(&FQ, &rr, &S2, &S1);
    if (fe_is_zero(&H)) {
        if (fe_is_zero(&rr)) { g1j_dbl(o, p); return; }
        g1j_set_inf(o);
        return;
    }
    f_add(&FQ, &rr, &rr, &rr);
    f_add(&FQ, &I, &H, &H);
    f_sqr(&FQ, &I, &I);
    f_mul(&FQ, &J, &H, &I);
    f_mul(&FQ, &V, &U1, &I);
    f_sqr(&FQ, &o->X, &rr);
    f_sub(&FQ, &o->X, &o->X, &J);
    f_sub(&FQ, &o->X, &o->X, &V);
    f_sub(&FQ, &o->X, &o->X, &V);
    f_sub(&FQ, &t, &V, &o->X);
    f_mul(&FQ, &t, &rr, &t);
    f_mul(&FQ, &J, &S1, &J);
    f_add(&FQ, &J, &J, &J);
    fe Ynew;
    f_sub(&FQ, &Ynew, &t, &J);
    f_add(&FQ, &t, &p->Z, &q->Z);
    f_sqr(&FQ, &t, &t);
    f_sub(&FQ, &t, &t, &Z1Z1);
    f_sub(&FQ, &t, &t, &Z2Z2);
    f_mul(&FQ, &o->Z, &t, &H);
    o->Y = Ynew;
}
static void g1j_mul(g1j* o, const g1a* p, const fe* k_canon) {
    g1j acc;
    g1j_set_inf(&acc);
    g1j base;
    g1j_from_affine(&base, p);
    for (int i = 0; i < 256; i++) {
        if ((k_canon->l[i >> 6] >> (i & 63)) & 1) {
            g1j t = acc;
            g1j_add(&acc, &t, &base);
        }
        g1j t = base;
        g1j_dbl(&base, &t);
    }
    *o = acc;
}


/* ------------------------------------------------------------ G2 (Fq2) */
/* BN254 G2: y^2 = x^3 + b2 over Fq2 = Fq[u]/(u^2+1), b2 = 3/(9+u) (D-twist;
 * halo2curves bn256::G2, SURVEY.md §8a minor row — sizes <= 2 in the
 * reference: SRS/verifier-side algebra only, so the oracle covers it and no
 * GPU kernel exists). Affine memory image (G2Affine): x.c0||x.c1||y.c0||y.c1,
 * 32 B LE Montgomery each; identity = 128 zero bytes. Affine chord/tangent
 * formulas with Fermat inversion (sizes are tiny). */
typedef struct { fe c0, c1; } fq2;
typedef struct { fq2 x, y; int inf; } g2a;

static void fq2_add(fq2* o, const fq2* a, const fq2* b) {
    f_add(&FQ, &o->c0, &a->c0, &b->c0);
    f_add(&FQ, &o->c1, &a->c1, &b->c1);
}
static void fq2_sub(fq2* o, const fq2* a, const fq2* b) {
    f_sub(&FQ, &o->c0, &a->c0, &b->c0);
    f_sub(&FQ, &o->c1, &a->c1, &b->c1);
}
static void fq2_neg(fq2* o, const fq2* a) {
    f_neg(&FQ, &o->c0, &a->c0);
    f_neg(&FQ, &o->c1, &a->c1);
}
static void fq2_mul(fq2* o, const fq2* a, const fq2* b) {
    fe t0, t1, t2, t3;
    f_mul(&FQ, &t0, &a->c0, &b->c0);
    f_mul(&FQ, &t1, &a->c1, &b->c1);
    f_mul(&FQ, &t2, &a->c0, &b->c1);
    f_mul(&FQ, &t3, &a->c1, &b->c0);
    f_sub(&FQ, &o->c0, &t0, &t1);   /* u^2 = -1 */
    f_add(&FQ, &o->c1, &t2, &t3);
}
static void fq2_inv(fq2* o, const fq2* a) {
    /* (c0 + c1 u)^-1 = (c0 - c1 u) / (c0^2 + c1^2) */
    fe n0, n1, d;
    f_sqr(&FQ, &n0, &a->c0);
    f_sqr(&FQ, &n1, &a->c1);
    f_add(&FQ, &d, &n0, &n1);
    f_inv(&FQ, &d, &d);
    f_mul(&FQ, &o->c0, &a->c0, &d);
    fe nc1;
    f_neg(&FQ, &nc1, &a->c1);
    f_mul(&FQ, &o->c1, &nc1, &d);
}
static int fq2_is_zero(const fq2* a) {
    return fe_is_zero(&a->c0) && fe_is_zero(&a->c1);
}

static void g2a_from_bytes(g2a* p, const uint8_t b[128]) {
    fe_from_bytes(&p->x.c0, b);
    fe_from_bytes(&p->x.c1, b + 32);
    fe_from_bytes(&p->y.c0, b + 64);
    fe_from_bytes(&p->y.c1, b + 96);
    p->inf = fq2_is_zero(&p->x) && fq2_is_zero(&p->y);
}
static void g2a_to_bytes(uint8_t b[128], const g2a* p) {
    if (p->inf) { memset(b, 0, 128); return; }
    fe_to_bytes(b, &p->x.c0);
    fe_to_bytes(b + 32, &p->x.c1);
    fe_to_bytes(b + 64, &p->y.c0);
    fe_to_bytes(b + 96, &p->y.c1);
}
static void g2a_add(g2a* o, const g2a* a, const g2a* b) {
    if (a->inf) { *o = *b; return; }
    if (b->inf) { *o = *a; return; }
    fq2 lam, t;
    if (fe_eq(&a->x.c0, &b->x.c0) && fe_eq(&a->x.c1, &b->x.c1)) {
        fq2 ysum;
        fq2_add(&ysum, &a->y, &b->y);
        if (fq2_is_zero(&ysum)) { memset(o, 0, sizeof *o); o->inf = 1; return; }
        /* tangent: 3 x^2 / 2y */
        fq2 num, den;
        fq2_mul(&num, &a->x, &a->x);
        fq2_add(&t, &num, &num);
        fq2_add(&num, &t, &num);
        fq2_add(&den, &a->y, &a->y);
        fq2_inv(&den, &den);
        fq2_mul(&lam, &num, &den);
    } else {
        fq2 num, den;
        fq2_sub(&num, &b->y, &a->y);
        fq2_sub(&den, &b->x, &a->x);
        fq2_inv(&den, &den);
        fq2_mul(&lam, &num, &den);
    }
    g2a r;
    r.inf = 0;
    fq2_mul(&r.x, &lam, &lam);
    fq2_sub(&r.x, &r.x, &a->x);
    fq2_sub(&r.x, &r.x, &b->x);
    fq2_sub(&t, &a->x, &r.x);
    fq2_mul(&r.y, &lam, &t);
    fq2_sub(&r.y, &r.y, &a->y);
    *o = r;
}
static void g2a_mul(g2a* o, const g2a* p, const fe* k_canon) {
    g2a acc, base = *p;
    memset(&acc, 0, sizeof acc);
    acc.inf = 1;
    for (int i = 0; i < 256; i++) {
        if ((k_canon->l[i >> 6] >> (i & 63)) & 1) {
            g2a t = acc;
            g2a_add(&acc, &t, &base);
        }
        g2a t = base;
        g2a_add(&base, &t, &t);
    }
    *o = acc;
}

void oracle_g2_add(const uint8_t a[128], const uint8_t b[128], uint8_t out[128]) {
    g2a A, B, O;
    g2a_from_bytes(&A, a);
    g2a_from_bytes(&B, b);
    g2a_add(&O, &A, &B);
    g2a_to_bytes(out, &O);
}
void oracle_g2_neg(const uint8_t a[128], uint8_t out[128]) {
    g2a A;
    g2a_from_bytes(&A, a);
    if (!A.inf) fq2_neg(&A.y, &A.y);
    g2a_to_bytes(out, &A);
}
void oracle_g2_mul(const uint8_t p[128], const uint8_t k_canon[32], uint8_t out[128]) {
    g2a A, O;
    fe k;
    g2a_from_bytes(&A, p);
    fe_from_bytes(&k, k_canon);
    g2a_mul(&O, &A, &k);
    g2a_to_bytes(out, &O);
}
int oracle_g2_is_on_curve(const uint8_t p[128]) {
    g2a A;
    g2a_from_bytes(&A, p);
    if (A.inf) return 1;
    /* b2 = 3/(9+u), computed on the fly (values in Montgomery form) */
    fq2 nine_u, b2, three, rhs, lhs, t;
    nine_u.c0 = FQ.one;
    f_add(&FQ, &nine_u.c0, &nine_u.c0, &FQ.one);
    fe two = nine_u.c0;                       /* 2 */
    f_add(&FQ, &nine_u.c0, &two, &two);       /* 4 */
    f_add(&FQ, &nine_u.c0, &nine_u.c0, &nine_u.c0); /* 8 */
    f_add(&FQ, &nine_u.c0, &nine_u.c0, &FQ.one);    /* 9 */
    nine_u.c1 = FQ.one;
    fq2_inv(&b2, &nine_u);
    f_add(&FQ, &three.c0, &two, &FQ.one);
    three.c1 = FE_ZERO;
    fq2_mul(&b2, &b2, &three);
    fq2_mul(&t, &A.x, &A.x);
    fq2_mul(&rhs, &t, &A.x);
    fq2_add(&rhs, &rhs, &b2);
    fq2_mul(&lhs, &A.y, &A.y);
    return fe_eq(&lhs.c0, &rhs.c0) && fe_eq(&lhs.c1, &rhs.c1);
}
/* Sum_i scalars_i * bases_i over G2 — the reference's only G2 MSMs are the
 * size <= 2 SRS/verifier ones, so a simple serial sum is the
 * whole requirement. */
void oracle_msm_g2(const uint8_t* bases, const uint8_t* scalars_canon,
                   uint64_t n, uint8_t out[128]) {
    g2a acc;
    memset(&acc, 0, sizeof acc);
    acc.inf = 1;
    for (uint64_t i = 0; i < n; i++) {
        g2a p, t, prod;
        fe k;
        g2a_from_bytes(&p, bases + 128 * i);
        fe_from_bytes(&k, scalars_canon + 32 * i);
        g2a_mul(&prod, &p, &k);
        t = acc;
        g2a_add(&acc, &t, &prod);
    }
    g2a_to_bytes(out, &acc);
}

/* ------------------------------------------------------------ public: fields */
#define FIELD_WRAP(name, ctx, op)                                              \
    void name(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]) {     \
        fe A, B, O;                                                            \
        fe_from_bytes(&A, a); fe_from_bytes(&B, b);                            \
        op(&ctx, &O, &A, &B);                                                  \
        fe_to_bytes(out, &O);                                                  \
    }
FIELD_WRAP(oracle_fr_add, FR, f_add)
FIELD_WRAP(oracle_fr_sub, FR, f_sub)
FIELD_WRAP(oracle_fr_mul, FR, f_mul)
FIELD_WRAP(oracle_fq_add, FQ, f_add)
FIELD_WRAP(oracle_fq_sub, FQ, f_sub)
FIELD_WRAP(oracle_fq_mul, FQ, f_mul)
void oracle_fr_inv(const uint8_t a[32], uint8_t out[32]) {
    fe A, O; fe_from_bytes(&A, a); f_inv(&FR, &O, &A); fe_to_bytes(out, &O);
}
void oracle_fq_inv(const uint8_t a[32], uint8_t out[32]) {
    fe A, O; fe_from_bytes(&A, a); f_inv(&FQ, &O, &A); fe_to_bytes(out, &O);
}
void oracle_fr_pow(const uint8_t a[32], const uint8_t e_canon[32], uint8_t out[32]) {
    fe A, E, O;
    fe_from_bytes(&A, a); fe_from_bytes(&E, e_canon);
    f_pow(&FR, &O, &A, &E);
    fe_to_bytes(out, &O);
}
void oracle_fr_to_canonical(const uint8_t a[32], uint8_t out[32]) {
    fe A, O; fe_from_bytes(&A, a); f_from_mont(&FR, &O, &A); fe_to_bytes(out, &O);
}
void oracle_fr_from_canonical(const uint8_t a[32], uint8_t out[32]) {
    fe A, O; fe_from_bytes(&A, a); f_to_mont(&FR, &O, &A); fe_to_bytes(out, &O);
}

/* ------------------------------------------------------------ public: G1 */
void oracle_g1_add(const uint8_t a[64], const uint8_t b[64], uint8_t out[64]) {
    g1a A, B, O;
    g1a_from_bytes(&A, a); g1a_from_bytes(&B, b);
    g1j J, R;
    g1j_from_affine(&J, &A);
    g1j_add_affine(&R, &J, &B);
    g1j_to_affine(&O, &R);
    g1a_to_bytes(out, &O);
}
void oracle_g1_neg(const uint8_t a[64], uint8_t out[64]) {
    g1a A;
    g1a_from_bytes(&A, a);
    if (!A.inf) f_neg(&FQ, &A.y, &A.y);
    g1a_to_bytes(out, &A);
}
void oracle_g1_mul(const uint8_t p[64], const uint8_t k_canon[32], uint8_t out[64]) {
    g1a A, O;
    fe K;
    g1a_from_bytes(&A, p); fe_from_bytes(&K, k_canon);
    g1j R;
    g1j_mul(&R, &A, &K);
    g1j_to_affine(&O, &R);
    g1a_to_bytes(out, &O);
}
int oracle_g1_is_on_curve(const uint8_t p[64]) {
    g1a A;
    g1a_from_bytes(&A, p);
    if (A.inf) return 1;
    fe y2, x3, t;
    f_sqr(&FQ, &y2, &A.y);
    f_sqr(&FQ, &x3, &A.x);
    f_mul(&FQ, &x3, &x3, &A.x);
    /* b = 3 in Montgomery form = one+one+one */
    f_add(&FQ, &t, &FQ.one, &FQ.one);
    f_add(&FQ, &t, &t, &FQ.one);
    f_add(&FQ, &x3, &x3, &t);
    return fe_eq(&y2, &x3);
}

/* ------------------------------------------------------------ MSM (Pippenger) */
static int msm_window_bits(uint64_t n) {
    /* free choice (result is window-independent); roughly log2(n) */
    int c = 3;
    while ((1ull << (c + 2)) < n && c < 16) c++;
    return c;
}
void oracle_msm_g1(const uint8_t* bases, const uint8_t* scalars, uint64_t n,
                   int scalars_canonical, uint8_t out[64]) {
    if (n == 0) { memset(out, 0, 64); return; }
    const int c = msm_window_bits(n);
    const int nwin = (254 + c - 1) / c;
    g1a* pts = (g1a*)malloc(n * sizeof(g1a));
    fe* sc = (fe*)malloc(n * sizeof(fe));
    for (uint64_t i = 0; i < n; i++) {
        g1a_from_bytes(&pts[i], bases + 64 * i);
        fe_from_bytes(&sc[i], scalars + 32 * i);
        if (!scalars_canonical) f_from_mont(&FR, &sc[i], &sc[i]);
    }
    g1j* winsum = (g1j*)malloc(nwin * sizeof(g1j));
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 1)
#endif
    for (int w = 0; w < nwin; w++) {
        const int nbuckets = 1 << c;
        g1j* buckets = (g1j*)malloc(nbuckets * sizeof(g1j));
        for (int b = 0; b < nbuckets; b++) g1j_set_inf(&buckets[b]);
        const int bit0 = w * c;
        for (uint64_t i = 0; i < n; i++) {
            /* extract c bits starting at bit0 from the canonical scalar */
            uint32_t d = 0;
            for (int b = 0; b < c; b++) {
                int bit = bit0 + b;
                if (bit < 256 && ((sc[i].l[bit >> 6] >> (bit & 63)) & 1)) d |= 1u << b;
            }
            if (d) {
                g1j t = buckets[d];
                g1j_add_affine(&buckets[d], &t, &pts[i]);
            }
        }
        /* sum_b b*bucket[b] via running suffix sums */
        g1j acc, total;
        g1j_set_inf(&acc);
        g1j_set_inf(&total);
        for (int b = nbuckets - 1; b >= 1; b--) {
            g1j t = acc;
            g1j_add(&acc, &t, &buckets[b]);
            t = total;
            g1j_add(&total, &t, &acc);
        }
        winsum[w] = total;
        free(buckets);
    }
    /* Horner over windows */
    g1j res;
    g1j_set_inf(&res);
    for (int w = nwin - 1; w >= 0; w--) {
        for (int d = 0; d < c && w != nwin - 1; d++) {
            g1j t = res;
            g1j_dbl(&res, &t);
        }
        g1j t = res;
        g1j_add(&res, &t, &winsum[w]);
    }
    g1a o;
    g1j_to_affine(&o, &res);
    g1a_to_bytes(out, &o);
    free(pts); free(sc); free(winsum);
}

/* ------------------------------------------------------------ NTT */
void oracle_ntt_fr(uint8_t* data, uint32_t log_n, const uint8_t omega[32],
                   int inverse, const uint8_t* coset_gen) {
    const uint64_t n = 1ull << log_n;
    fe* a = (fe*)data; /* memory image == limb array */
    fe om, g;
    fe_from_bytes(&om, omega);
    if (coset_gen) fe_from_bytes(&g, (const uint8_t*)coset_gen);
    if (coset_gen && !inverse) {
        fe cur = FR.one;
        for (uint64_t i = 0; i < n; i++) {
            f_mul(&FR, &a[i], &a[i], &cur);
            f_mul(&FR, &cur, &cur, &g);
        }
    }
    /* bit-reversal permutation */
    for (uint64_t i = 0; i < n; i++) {
        uint64_t j = 0;
        for (uint32_t b = 0; b < log_n; b++) j |= ((i >> b) & 1) << (log_n - 1 - b);
        if (j > i) { fe t = a[i]; a[i] = a[j]; a[j] = t; }
    }
    /* twiddle table: tw[j] = omega^j for j < n/2 */
    uint64_t half = n >> 1;
    fe* tw = NULL;
    if (log_n > 0) {
        tw = (fe*)malloc((half ? half : 1) * sizeof(fe));
        tw[0] = FR.one;
        for (uint64_t j = 1; j < half; j++) f_mul(&FR, &tw[j], &tw[j - 1], &om);
    }
    for (uint32_t s = 1; s <= log_n; s++) {
        const uint64_t m = 1ull << s;
        const uint64_t tstride = n / m; /* omega_m^j = omega^(j * n/m) */
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
        for (uint64_t k = 0; k < n; k += m) {
            for (uint64_t j = 0; j < m / 2; j++) {
                fe t, u;
                f_mul(&FR, &t, &tw[j * tstride], &a[k + j + m / 2]);
                u = a[k + j];
                f_add(&FR, &a[k + j], &u, &t);
                f_sub(&FR, &a[k + j + m / 2], &u, &t);
            }
        }
    }
    free(tw);
    if (inverse) {
        /* n^{-1} in Montgomery form */
        fe ncanon = {{0, 0, 0, 0}};
        ncanon.l[log_n >> 6] = 1ull << (log_n & 63);
        fe nm, ninv;
        f_to_mont(&FR, &nm, &ncanon);
        f_inv(&FR, &ninv, &nm);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
        for (uint64_t i = 0; i < n; i++) f_mul(&FR, &a[i], &a[i], &ninv);
    }
    if (coset_gen && inverse) {
        fe cur = FR.one;
        for (uint64_t i = 0; i < n; i++) {
            f_mul(&FR, &a[i], &a[i], &cur);
            f_mul(&FR, &cur, &cur, &g);
        }
    }
}

/* ------------------------------------------------------- input generation */
typedef struct { uint64_t z; } smix;
static uint64_t smix_next(smix* s) {
    s->z += 0x9E3779B97F4A7C15ull;
    uint64_t x = s->z;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}
/* 256-bit draw reduced mod m -> canonical fe (schoolbook mod via Montgomery:
 * v mod m = from_mont(to_mont(v)); to_mont needs v < m? No: CIOS handles any
 * 256-bit input < 2^256 since it reduces mod m as it goes. */
static void smix_fe(smix* s, const fctx* f, fe* o) {
    fe v;
    for (int i = 0; i < 4; i++) v.l[i] = smix_next(s);
    /* reduce: t = v * R mod m (Montgomery mul by R2 gives v*R; then back) */
    fe t;
    f_mul(f, &t, &v, &f->r2);   /* t = v * R mod m */
    f_from_mont(f, o, &t);      /* o = v mod m (canonical) */
}
static const g1a G1_GEN_A = {
    .x = {{0xd35d438dc58f0d9dull, 0x0a78eb28f5c70b3dull, 0x666ea36f7879462cull, 0x0e0a77c19a07df2full}},
    .y = {{0xa6ba871b8b1e1b3aull, 0x14f1d651eb8e167bull, 0xccdd46def0f28c58ull, 0x1c14ef83340fbe5eull}},
    .inf = 0,
};
void oracle_gen_fr_vector(uint64_t n, uint64_t seed, uint8_t* out_mont) {
    smix s = {seed};
    for (uint64_t i = 0; i < n; i++) {
        fe c, m;
        smix_fe(&s, &FR, &c);
        f_to_mont(&FR, &m, &c);
        fe_to_bytes(out_mont + 32 * i, &m);
    }
}
void oracle_gen_msm_inputs(uint64_t n, uint64_t seed,
                           uint8_t* scalars_canon, uint8_t* bases) {
    smix s = {seed};
    for (uint64_t i = 0; i < n; i++) {
        fe sc, k;
        smix_fe(&s, &FR, &sc);
        fe_to_bytes(scalars_canon + 32 * i, &sc);
        smix_fe(&s, &FR, &k);
        g1j R;
        g1a O;
        g1j_mul(&R, &G1_GEN_A, &k);
        g1j_to_affine(&O, &R);
        g1a_to_bytes(bases + 64 * i, &O);
    }
}
void oracle_gen_msm_inputs_fast(uint64_t n, uint64_t seed,
                                uint8_t* scalars_canon, uint8_t* bases) {
    smix s = {seed};
    for (uint64_t i = 0; i < n; i++) {
        fe sc;
        smix_fe(&s, &FR, &sc);
        fe_to_bytes(scalars_canon + 32 * i, &sc);
    }
    if (n == 0) return;
    fe k;
    smix_fe(&s, &FR, &k);
    g1j* chain = (g1j*)malloc(n * sizeof(g1j));
    g1j_mul(&chain[0], &G1_GEN_A, &k);
    for (uint64_t i = 1; i < n; i++) g1j_add_affine(&chain[i], &chain[i - 1], &G1_GEN_A);
    /* batch-normalize (Montgomery's trick) */
    fe* pref = (fe*)malloc(n * sizeof(fe));
    fe acc = FQ.one;
    for (uint64_t i = 0; i < n; i++) {
        pref[i] = acc;                      /* product of Z_0..Z_{i-1} */
        f_mul(&FQ, &acc, &acc, &chain[i].Z);
    }
    fe accinv;
    f_inv(&FQ, &accinv, &acc);
    for (uint64_t ii = n; ii-- > 0;) {
        fe zi;
        f_mul(&FQ, &zi, &accinv, &pref[ii]);       /* Z_ii^{-1} */
        f_mul(&FQ, &accinv, &accinv, &chain[ii].Z);
        fe zi2, zi3;
        f_sqr(&FQ, &zi2, &zi);
        f_mul(&FQ, &zi3, &zi2, &zi);
        g1a o;
        f_mul(&FQ, &o.x, &chain[ii].X, &zi2);
        f_mul(&FQ, &o.y, &chain[ii].Y, &zi3);
        o.inf = 0;
        g1a_to_bytes(bases + 64 * ii, &o);
    }
    free(chain); free(pref);
}
int oracle_num_threads(void) {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}
