/* ============================================================================
 * CPU ORACLE for BLS12-381 G1 — test infrastructure ONLY (see
 * bn254_oracle.c header for the oracle discipline).
 *
 * Semantics restated from the reference's in-tree blst path
 * (crates/common/crypto/bls_blst.rs, the EIP-2537 backend used by
 * crates/common/crypto/provider.rs:592-668, and the KZG MSM call path
 * crates/common/crypto/kzg.rs:208-230):
 *  - coordinates: 48-byte big-endian unpadded, CANONICAL (>= p rejected —
 *    bls_blst.rs MODULUS_REPR check; stricter than the bn254 ark path);
 *  - (0,0) = point at infinity on input and output;
 *  - on-curve check (y^2 = x^3 + 4); G1 MSM additionally enforces the
 *    r-subgroup check (read_g1_subgroup, bls_blst.rs:215-222);
 *  - scalars: 32-byte big-endian used as FULL 256-bit integers
 *    (SCALAR_BITS = 256, no reduction — bls_blst.rs:33,103).
 *
 * Field: 6x64-limb Montgomery, R = 2^384 (independent of the GPU's 14x29
 * representation).  Curve: y^2 = x^3 + 4, standard G1 generator.
 * ==========================================================================*/

#include <stdint.h>
#include <stddef.h>
#include <string.h>
#include <stdlib.h>

#include "bn254_constants.h"

#ifdef _OPENMP
#include <omp.h>
#endif

typedef unsigned __int128 u128;

#define NL 6 /* limbs */

typedef struct { uint64_t v[NL]; } fb; /* Fp element, 6x64 LE limbs */

static int fb_cmp(const fb *a, const fb *b) {
    for (int i = NL - 1; i >= 0; i--) {
        if (a->v[i] < b->v[i]) return -1;
        if (a->v[i] > b->v[i]) return 1;
    }
    return 0;
}

static int fb_is_zero(const fb *a) {
    uint64_t d = 0;
    for (int i = 0; i < NL; i++) d |= a->v[i];
    return d == 0;
}

static void fb_from_limbs(fb *o, const uint64_t *l) { memcpy(o->v, l, 8 * NL); }

static uint64_t fb_add6(fb *o, const fb *a, const fb *b) {
    u128 c = 0;
    for (int i = 0; i < NL; i++) {
        c += (u128)a->v[i] + b->v[i];
        o->v[i] = (uint64_t)c;
        c >>= 64;
    }
    return (uint64_t)c;
}

static uint64_t fb_sub6(fb *o, const fb *a, const fb *b) {
    u128 bor = 0;
    for (int i = 0; i < NL; i++) {
        u128 t = (u128)a->v[i] - b->v[i] - bor;
        o->v[i] = (uint64_t)t;
        bor = (t >> 64) & 1;
    }
    return (uint64_t)bor;
}

static void fb_cond_sub(fb *a, uint64_t carry) {
    fb m, t;
    fb_from_limbs(&m, BLSP_MOD);
    if (carry || fb_cmp(a, &m) >= 0) {
        fb_sub6(&t, a, &m);
        *a = t;
    }
}

static void fb_mod_add(fb *o, const fb *a, const fb *b) {
    uint64_t c = fb_add6(o, a, b);
    fb_cond_sub(o, c);
}

static void fb_mod_sub(fb *o, const fb *a, const fb *b) {
    if (fb_sub6(o, a, b)) {
        fb m, t;
        fb_from_limbs(&m, BLSP_MOD);
        fb_add6(&t, o, &m);
        *o = t;
    }
}

/* SOS Montgomery multiplication, 6 limbs */
static void fb_mont_mul(fb *o, const fb *a, const fb *b) {
    uint64_t t[2 * NL + 1] = {0};
    for (int i = 0; i < NL; i++) {
        u128 c = 0;
        for (int j = 0; j < NL; j++) {
            c += (u128)a->v[i] * b->v[j] + t[i + j];
            t[i + j] = (uint64_t)c;
            c >>= 64;
        }
        for (int k = i + NL; c && k < 2 * NL + 1; k++) {
            c += t[k];
            t[k] = (uint64_t)c;
            c >>= 64;
        }
    }
    for (int i = 0; i < NL; i++) {
        uint64_t m = t[i] * BLSP_N0INV;
        u128 c = 0;
        for (int j = 0; j < NL; j++) {
            c += (u128)m * BLSP_MOD[j] + t[i + j];
            t[i + j] = (uint64_t)c;
            c >>= 64;
        }
        for (int k = i + NL; c && k < 2 * NL + 1; k++) {
            c += t[k];
            t[k] = (uint64_t)c;
            c >>= 64;
        }
    }
    fb r;
    memcpy(r.v, t + NL, 8 * NL);
    fb_cond_sub(&r, t[2 * NL]);
    *o = r;
}

static void fb_sqr(fb *o, const fb *a) { fb_mont_mul(o, a, a); }

static void fb_to_mont(fb *o, const fb *x) {
    fb r2;
    fb_from_limbs(&r2, BLSP_R2);
    fb_mont_mul(o, x, &r2);
}

static void fb_from_mont(fb *o, const fb *x) {
    fb one = {{1, 0, 0, 0, 0, 0}};
    fb_mont_mul(o, x, &one);
}

static void fb_pow(fb *o, const fb *x, const fb *e) {
    fb acc, base = *x;
    fb_from_limbs(&acc, BLSP_R);
    for (int i = 64 * NL - 1; i >= 0; i--) {
        fb_sqr(&acc, &acc);
        if ((e->v[i / 64] >> (i % 64)) & 1) fb_mont_mul(&acc, &acc, &base);
    }
    *o = acc;
}

static void fb_inv(fb *o, const fb *x) {
    fb e, m, two = {{2, 0, 0, 0, 0, 0}};
    fb_from_limbs(&m, BLSP_MOD);
    fb_sub6(&e, &m, &two);
    fb_pow(o, x, &e);
}

/* 48-byte big-endian <-> limbs */
static void fb_from_be(fb *o, const uint8_t *b) {
    for (int i = 0; i < NL; i++) {
        uint64_t v = 0;
        for (int j = 0; j < 8; j++) v = (v << 8) | b[(NL - 1 - i) * 8 + j];
        o->v[i] = v;
    }
}

static void fb_to_be(uint8_t *b, const fb *x) {
    for (int i = 0; i < NL; i++)
        for (int j = 0; j < 8; j++)
            b[(NL - 1 - i) * 8 + j] = (uint8_t)(x->v[i] >> (56 - 8 * j));
}

/* ---------------- G1 (Jacobian, Montgomery) ---------------- */

typedef struct { fb x, y, z; } bg1j;
typedef struct { fb x, y; } bg1a;

static void bg1_set_inf(bg1j *p) {
    fb_from_limbs(&p->x, BLSP_R);
    fb_from_limbs(&p->y, BLSP_R);
    memset(&p->z, 0, sizeof(fb));
}

static int bg1_is_inf(const bg1j *p) { return fb_is_zero(&p->z); }

static void bg1_dbl(bg1j *o, const bg1j *p) {
    if (bg1_is_inf(p)) { *o = *p; return; }
    fb A, B, C, D, E, F, t, t2;
    fb_sqr(&A, &p->x);
    fb_sqr(&B, &p->y);
    fb_sqr(&C, &B);
    fb_mod_add(&t, &p->x, &B);
    fb_sqr(&t, &t);
    fb_mod_sub(&t, &t, &A);
    fb_mod_sub(&t, &t, &C);
    fb_mod_add(&D, &t, &t);
    fb_mod_add(&E, &A, &A);
    fb_mod_add(&E, &E, &A);
    fb_sqr(&F, &E);
    fb_mod_sub(&t, &F, &D);
    fb_mod_sub(&o->x, &t, &D);
    fb_mod_sub(&t, &D, &o->x);
    fb_mont_mul(&t, &E, &t);
    fb_mod_add(&t2, &C, &C);
    fb_mod_add(&t2, &t2, &t2);
    fb_mod_add(&t2, &t2, &t2);
    fb y3, z3;
    fb_mod_sub(&y3, &t, &t2);
    fb_mont_mul(&z3, &p->y, &p->z);
    fb_mod_add(&z3, &z3, &z3);
    o->y = y3;
    o->z = z3;
}

static void bg1_add(bg1j *o, const bg1j *p, const bg1j *q) {
    if (bg1_is_inf(p)) { *o = *q; return; }
    if (bg1_is_inf(q)) { *o = *p; return; }
    fb z1z1, z2z2, u1, u2, s1, s2, h, r, t;
    fb_sqr(&z1z1, &p->z);
    fb_sqr(&z2z2, &q->z);
    fb_mont_mul(&u1, &p->x, &z2z2);
    fb_mont_mul(&u2, &q->x, &z1z1);
    fb_mont_mul(&t, &q->z, &z2z2);
    fb_mont_mul(&s1, &p->y, &t);
    fb_mont_mul(&t, &p->z, &z1z1);
    fb_mont_mul(&s2, &q->y, &t);
    fb_mod_sub(&h, &u2, &u1);
    fb_mod_sub(&r, &s2, &s1);
    if (fb_is_zero(&h)) {
        if (fb_is_zero(&r)) { bg1_dbl(o, p); return; }
        bg1_set_inf(o);
        return;
    }
    fb hh, hhh, v;
    fb_sqr(&hh, &h);
    fb_mont_mul(&hhh, &h, &hh);
    fb_mont_mul(&v, &u1, &hh);
    fb x3, y3, z3;
    fb_sqr(&x3, &r);
    fb_mod_sub(&x3, &x3, &hhh);
    fb_mod_sub(&x3, &x3, &v);
    fb_mod_sub(&x3, &x3, &v);
    fb_mod_sub(&t, &v, &x3);
    fb_mont_mul(&y3, &r, &t);
    fb_mont_mul(&t, &s1, &hhh);
    fb_mod_sub(&y3, &y3, &t);
    fb_mont_mul(&z3, &p->z, &q->z);
    fb_mont_mul(&z3, &z3, &h);
    o->x = x3;
    o->y = y3;
    o->z = z3;
}

static void bg1_add_affine(bg1j *o, const bg1j *p, const bg1a *q) {
    if (bg1_is_inf(p)) {
        o->x = q->x;
        o->y = q->y;
        fb_from_limbs(&o->z, BLSP_R);
        return;
    }
    fb z1z1, u2, s2, h, r, t;
    fb_sqr(&z1z1, &p->z);
    fb_mont_mul(&u2, &q->x, &z1z1);
    fb_mont_mul(&t, &p->z, &z1z1);
    fb_mont_mul(&s2, &q->y, &t);
    fb_mod_sub(&h, &u2, &p->x);
    fb_mod_sub(&r, &s2, &p->y);
    if (fb_is_zero(&h)) {
        if (fb_is_zero(&r)) { bg1_dbl(o, p); return; }
        bg1_set_inf(o);
        return;
    }
    fb hh, hhh, v;
    fb_sqr(&hh, &h);
    fb_mont_mul(&hhh, &h, &hh);
    fb_mont_mul(&v, &p->x, &hh);
    fb x3, y3, z3;
    fb_sqr(&x3, &r);
    fb_mod_sub(&x3, &x3, &hhh);
    fb_mod_sub(&x3, &x3, &v);
    fb_mod_sub(&x3, &x3, &v);
    fb_mod_sub(&t, &v, &x3);
    fb_mont_mul(&y3, &r, &t);
    fb_mont_mul(&t, &p->y, &hhh);
    fb_mod_sub(&y3, &y3, &t);
    fb_mont_mul(&z3, &p->z, &h);
    o->x = x3;
    o->y = y3;
    o->z = z3;
}

static int bg1a_on_curve(const bg1a *p) {
    fb l, r, b4;
    fb_sqr(&l, &p->y);
    fb_sqr(&r, &p->x);
    fb_mont_mul(&r, &r, &p->x);
    fb_from_limbs(&b4, BLS_B4_MONT);
    fb_mod_add(&r, &r, &b4);
    return fb_cmp(&l, &r) == 0;
}

static void bg1_to_affine_be(uint8_t out[96], const bg1j *p) {
    if (bg1_is_inf(p)) {
        memset(out, 0, 96);
        return;
    }
    fb zi, zi2, zi3, xa, ya, xc, yc;
    fb_inv(&zi, &p->z);
    fb_sqr(&zi2, &zi);
    fb_mont_mul(&zi3, &zi2, &zi);
    fb_mont_mul(&xa, &p->x, &zi2);
    fb_mont_mul(&ya, &p->y, &zi3);
    fb_from_mont(&xc, &xa);
    fb_from_mont(&yc, &ya);
    fb_to_be(out, &xc);
    fb_to_be(out + 48, &yc);
}

#define BORC_OK 0
#define BORC_ERR_POINT 1
#define BORC_ERR_INPUT 2
#define BORC_ERR_SUBGROUP 3

/* parse 96-byte BE affine per EIP-2537/blst semantics: coords CANONICAL
 * (>= p rejected), (0,0) -> infinity, on-curve check. */
static int bg1_parse_be(bg1a *o, int *is_inf, const uint8_t in[96]) {
    fb x, y, m;
    fb_from_be(&x, in);
    fb_from_be(&y, in + 48);
    fb_from_limbs(&m, BLSP_MOD);
    if (fb_cmp(&x, &m) >= 0 || fb_cmp(&y, &m) >= 0) return BORC_ERR_INPUT;
    if (fb_is_zero(&x) && fb_is_zero(&y)) {
        *is_inf = 1;
        return BORC_OK;
    }
    fb_to_mont(&o->x, &x);
    fb_to_mont(&o->y, &y);
    *is_inf = 0;
    if (!bg1a_on_curve(o)) return BORC_ERR_POINT;
    return BORC_OK;
}

/* full 256-bit scalar mul (blst SCALAR_BITS=256: no reduction) */
static void bg1_scalar_mul_be(bg1j *o, const bg1a *p, const uint8_t k[32]) {
    bg1j acc;
    bg1_set_inf(&acc);
    for (int i = 0; i < 256; i++) {
        bg1_dbl(&acc, &acc);
        if ((k[i / 8] >> (7 - (i % 8))) & 1) bg1_add_affine(&acc, &acc, p);
    }
    *o = acc;
}

/* subgroup check: r*P == infinity (read_g1_subgroup semantics) */
static int bg1_in_subgroup(const bg1a *p) {
    uint8_t rbe[32];
    for (int i = 0; i < 4; i++)
        for (int j = 0; j < 8; j++)
            rbe[(3 - i) * 8 + j] = (uint8_t)(BLSR_MOD[i] >> (56 - 8 * j));
    bg1j t;
    bg1_scalar_mul_be(&t, p, rbe);
    return bg1_is_inf(&t);
}

/* =================== public oracle API =================== */

int oracle_bls_g1_add(const uint8_t p1[96], const uint8_t p2[96],
                      uint8_t out[96]) {
    bg1a a, b;
    int ia, ib, rc;
    if ((rc = bg1_parse_be(&a, &ia, p1))) return rc;
    if ((rc = bg1_parse_be(&b, &ib, p2))) return rc;
    bg1j acc;
    bg1_set_inf(&acc);
    if (!ia) bg1_add_affine(&acc, &acc, &a);
    if (!ib) bg1_add_affine(&acc, &acc, &b);
    bg1_to_affine_be(out, &acc);
    return BORC_OK;
}

int oracle_bls_g1_mul(const uint8_t point[96], const uint8_t scalar[32],
                      uint8_t out[96]) {
    bg1a p;
    int inf, rc;
    if ((rc = bg1_parse_be(&p, &inf, point))) return rc;
    if (inf) { memset(out, 0, 96); return BORC_OK; }
    bg1j r;
    bg1_scalar_mul_be(&r, &p, scalar);
    bg1_to_affine_be(out, &r);
    return BORC_OK;
}

/* Pippenger MSM, c=16 over the raw 256-bit scalars (16 windows);
 * subgroup check per point (bls_blst.rs g1_msm -> read_g1_subgroup). */
#define BC 16
#define BNWIN 16
#define BNBUCKET ((1u << BC) - 1)

static int bls_msm_core(const uint8_t *points, const uint8_t *scalars, size_t n,
                        bg1j *result) {
    bg1a *pts = malloc(n * sizeof(bg1a));
    uint8_t *inf = malloc(n);
    int err = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 16)
#endif
    for (size_t i = 0; i < n; i++) {
        int ii, rc;
        if ((rc = bg1_parse_be(&pts[i], &ii, points + 96 * i))) {
#ifdef _OPENMP
#pragma omp atomic write
#endif
            err = rc;
            continue;
        }
        inf[i] = (uint8_t)ii;
        if (!ii && !bg1_in_subgroup(&pts[i])) {
#ifdef _OPENMP
#pragma omp atomic write
#endif
            err = BORC_ERR_SUBGROUP;
        }
    }
    if (err) { free(pts); free(inf); return err; }

    bg1j wsum[BNWIN];
#ifdef _OPENMP
#pragma omp parallel
#endif
    {
        bg1j *buckets = malloc(BNBUCKET * sizeof(bg1j));
#ifdef _OPENMP
#pragma omp for schedule(dynamic)
#endif
        for (int w = 0; w < BNWIN; w++) {
            for (uint32_t b = 0; b < BNBUCKET; b++) bg1_set_inf(&buckets[b]);
            for (size_t i = 0; i < n; i++) {
                if (inf[i]) continue;
                /* window w = bits [16w, 16w+16) of the BE scalar */
                const uint8_t *k = scalars + 32 * i;
                int byte_hi = 31 - (2 * w + 1), byte_lo = 31 - 2 * w;
                uint32_t d = ((uint32_t)k[byte_hi] << 8) | k[byte_lo];
                if (d == 0) continue;
                bg1_add_affine(&buckets[d - 1], &buckets[d - 1], &pts[i]);
            }
            bg1j run, tot;
            bg1_set_inf(&run);
            bg1_set_inf(&tot);
            for (int64_t d = BNBUCKET - 1; d >= 0; d--) {
                bg1_add(&run, &run, &buckets[d]);
                bg1_add(&tot, &tot, &run);
            }
            wsum[w] = tot;
        }
        free(buckets);
    }
    bg1j acc = wsum[BNWIN - 1];
    for (int w = BNWIN - 2; w >= 0; w--) {
        for (int d = 0; d < BC; d++) bg1_dbl(&acc, &acc);
        bg1_add(&acc, &acc, &wsum[w]);
    }
    *result = acc;
    free(pts);
    free(inf);
    return BORC_OK;
}

int oracle_bls_g1_msm(const uint8_t *points, const uint8_t *scalars, size_t n,
                      uint8_t out[96]) {
    bg1j r;
    int rc = bls_msm_core(points, scalars, n, &r);
    if (rc) return rc;
    bg1_to_affine_be(out, &r);
    return BORC_OK;
}

int oracle_bls_g1_msm_naive(const uint8_t *points, const uint8_t *scalars,
                            size_t n, uint8_t out[96]) {
    bg1j acc;
    bg1_set_inf(&acc);
    for (size_t i = 0; i < n; i++) {
        bg1a p;
        int inf, rc;
        if ((rc = bg1_parse_be(&p, &inf, points + 96 * i))) return rc;
        if (inf) continue;
        if (!bg1_in_subgroup(&p)) return BORC_ERR_SUBGROUP;
        bg1j t;
        bg1_scalar_mul_be(&t, &p, scalars + 32 * i);
        bg1_add(&acc, &acc, &t);
    }
    bg1_to_affine_be(out, &acc);
    return BORC_OK;
}

/* Jacobian partial (144 bytes X||Y||Z BE canonical; Z=0 = infinity) */
int oracle_bls_g1_msm_jacobian(const uint8_t *points, const uint8_t *scalars,
                               size_t n, uint8_t out[144]) {
    bg1j r;
    int rc = bls_msm_core(points, scalars, n, &r);
    if (rc) return rc;
    fb xc, yc, zc;
    fb_from_mont(&xc, &r.x);
    fb_from_mont(&yc, &r.y);
    fb_from_mont(&zc, &r.z);
    fb_to_be(out, &xc);
    fb_to_be(out + 48, &yc);
    fb_to_be(out + 96, &zc);
    return BORC_OK;
}

int oracle_bls_g1_combine_jacobian(const uint8_t *jac, size_t g,
                                   uint8_t out[96]) {
    bg1j acc;
    bg1_set_inf(&acc);
    for (size_t i = 0; i < g; i++) {
        bg1j p;
        fb x, y, z;
        fb_from_be(&x, jac + 144 * i);
        fb_from_be(&y, jac + 144 * i + 48);
        fb_from_be(&z, jac + 144 * i + 96);
        fb_to_mont(&p.x, &x);
        fb_to_mont(&p.y, &y);
        fb_to_mont(&p.z, &z);
        bg1_add(&acc, &acc, &p);
    }
    bg1_to_affine_be(out, &acc);
    return BORC_OK;
}

/* P_i = (start+i+1)*G, 96-byte BE affine, via incremental adds + batch inv */
int oracle_bls_gen_points(uint64_t start, size_t n, uint8_t *out) {
    if (n == 0) return BORC_OK;
    bg1a gen;
    fb_from_limbs(&gen.x, BLS_GX_MONT);
    fb_from_limbs(&gen.y, BLS_GY_MONT);
    bg1j *acc = malloc(n * sizeof(bg1j));
    uint8_t k[32] = {0};
    uint64_t k0 = start + 1;
    for (int j = 0; j < 8; j++) k[31 - j] = (uint8_t)(k0 >> (8 * j));
    bg1_scalar_mul_be(&acc[0], &gen, k);
    for (size_t i = 1; i < n; i++) bg1_add_affine(&acc[i], &acc[i - 1], &gen);
    fb *pref = malloc((n + 1) * sizeof(fb));
    fb_from_limbs(&pref[0], BLSP_R);
    for (size_t i = 0; i < n; i++) fb_mont_mul(&pref[i + 1], &pref[i], &acc[i].z);
    fb inv_all;
    fb_inv(&inv_all, &pref[n]);
    for (size_t i = n; i-- > 0;) {
        fb zi, zi2, zi3, xa, ya, xc, yc;
        fb_mont_mul(&zi, &inv_all, &pref[i]);
        fb_mont_mul(&inv_all, &inv_all, &acc[i].z);
        fb_sqr(&zi2, &zi);
        fb_mont_mul(&zi3, &zi2, &zi);
        fb_mont_mul(&xa, &acc[i].x, &zi2);
        fb_mont_mul(&ya, &acc[i].y, &zi3);
        fb_from_mont(&xc, &xa);
        fb_from_mont(&yc, &ya);
        fb_to_be(out + 96 * i, &xc);
        fb_to_be(out + 96 * i + 48, &yc);
    }
    free(pref);
    free(acc);
    return BORC_OK;
}

/* n scalars uniform in [0, r_bls) (for KZG-style inputs where the blob
 * elements are canonical Fr), 32-byte BE; same xoshiro scheme as bn254. */
extern void oracle_gen_fr(uint64_t seed, size_t n, uint8_t *out); /* bn254 */

void oracle_bls_gen_fr(uint64_t seed, size_t n, uint8_t *out) {
    /* splitmix64-seeded xoshiro256++, 255-bit mask, reject >= r_bls */
    uint64_t s[4], sm = seed;
    for (int i = 0; i < 4; i++) {
        uint64_t z = (sm += 0x9e3779b97f4a7c15ull);
        z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
        z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
        s[i] = z ^ (z >> 31);
    }
    for (size_t i = 0; i < n; i++) {
        uint64_t v[4];
        int ok = 0;
        while (!ok) {
            for (int w = 0; w < 4; w++) {
                uint64_t r = ((s[0] + s[3]) << 23 | (s[0] + s[3]) >> 41) + s[0];
                uint64_t t = s[1] << 17;
                s[2] ^= s[0];
                s[3] ^= s[1];
                s[1] ^= s[2];
                s[0] ^= s[3];
                s[2] ^= t;
                s[3] = (s[3] << 45) | (s[3] >> 19);
                v[w] = r;
            }
            v[3] &= 0x7fffffffffffffffull; /* 255 bits */
            ok = 1;
            for (int w = 3; w >= 0; w--) {
                if (v[w] < BLSR_MOD[w]) break;
                if (v[w] > BLSR_MOD[w]) { ok = 0; break; }
                if (w == 0) ok = 0; /* equal */
            }
        }
        for (int w = 0; w < 4; w++)
            for (int j = 0; j < 8; j++)
                out[32 * i + (3 - w) * 8 + j] = (uint8_t)(v[w] >> (56 - 8 * j));
    }
}

/* ================= BLS12-381 G2: Fp2 = Fp[u]/(u^2+1) =================
 * EIP-2537 / blst semantics (crates/common/crypto/bls_blst.rs:226-255,
 * 303-321, 338-345, 395-441): 192-byte points x.c0||x.c1||y.c0||y.c1 with
 * canonical 48-B BE coords, (0,0,0,0) identity, on-curve for add,
 * on-curve + r-subgroup for MSM inputs; curve y^2 = x^3 + 4(1+u). */

typedef struct { fb c0, c1; } f2;
typedef struct { f2 x, y; } bg2a;
typedef struct { f2 x, y, z; } bg2j;

static void f2_add(f2 *o, const f2 *a, const f2 *b) {
    fb_mod_add(&o->c0, &a->c0, &b->c0);
    fb_mod_add(&o->c1, &a->c1, &b->c1);
}
static void f2_sub(f2 *o, const f2 *a, const f2 *b) {
    fb_mod_sub(&o->c0, &a->c0, &b->c0);
    fb_mod_sub(&o->c1, &a->c1, &b->c1);
}
static void f2_mul(f2 *o, const f2 *a, const f2 *b) {
    fb m0, m1, t0, t1, r0, r1;
    fb_mont_mul(&m0, &a->c0, &b->c0);
    fb_mont_mul(&m1, &a->c1, &b->c1);
    fb_mont_mul(&t0, &a->c0, &b->c1);
    fb_mont_mul(&t1, &a->c1, &b->c0);
    fb_mod_sub(&r0, &m0, &m1);
    fb_mod_add(&r1, &t0, &t1);
    o->c0 = r0;
    o->c1 = r1;
}
static void f2_sqr(f2 *o, const f2 *a) { f2_mul(o, a, a); }
static int f2_is_zero(const f2 *a) {
    return fb_is_zero(&a->c0) && fb_is_zero(&a->c1);
}
static int f2_eq(const f2 *a, const f2 *b) {
    return fb_cmp(&a->c0, &b->c0) == 0 && fb_cmp(&a->c1, &b->c1) == 0;
}
static void f2_inv(f2 *o, const f2 *a) {
    fb n0, n1, t, z;
    fb_sqr(&n0, &a->c0);
    fb_sqr(&n1, &a->c1);
    fb_mod_add(&n0, &n0, &n1);
    fb_inv(&t, &n0);
    memset(&z, 0, sizeof(fb));
    fb_mont_mul(&o->c0, &a->c0, &t);
    fb_mont_mul(&n1, &a->c1, &t);
    fb_mod_sub(&o->c1, &z, &n1);
}

static void bg2_set_inf(bg2j *p) {
    fb_from_limbs(&p->x.c0, BLSP_R);
    memset(&p->x.c1, 0, sizeof(fb));
    p->y = p->x;
    memset(&p->z, 0, sizeof(f2));
}
static int bg2_is_inf(const bg2j *p) { return f2_is_zero(&p->z); }

static void bg2_dbl(bg2j *o, const bg2j *p) {
    if (bg2_is_inf(p)) { *o = *p; return; }
    f2 A, B, C, D, E, F, t, t2, y3, z3;
    f2_sqr(&A, &p->x);
    f2_sqr(&B, &p->y);
    f2_sqr(&C, &B);
    f2_add(&t, &p->x, &B);
    f2_sqr(&t, &t);
    f2_sub(&t, &t, &A);
    f2_sub(&t, &t, &C);
    f2_add(&D, &t, &t);
    f2_add(&E, &A, &A);
    f2_add(&E, &E, &A);
    f2_sqr(&F, &E);
    f2_sub(&t, &F, &D);
    f2_sub(&o->x, &t, &D);
    f2_sub(&t, &D, &o->x);
    f2_mul(&t, &E, &t);
    f2_add(&t2, &C, &C);
    f2_add(&t2, &t2, &t2);
    f2_add(&t2, &t2, &t2);
    f2_sub(&y3, &t, &t2);
    f2_mul(&z3, &p->y, &p->z);
    f2_add(&z3, &z3, &z3);
    o->y = y3;
    o->z = z3;
}

static void bg2_add_affine(bg2j *o, const bg2j *p, const bg2a *q) {
    if (bg2_is_inf(p)) {
        o->x = q->x;
        o->y = q->y;
        fb_from_limbs(&o->z.c0, BLSP_R);
        memset(&o->z.c1, 0, sizeof(fb));
        return;
    }
    f2 z1z1, u2, s2, h, r, t;
    f2_sqr(&z1z1, &p->z);
    f2_mul(&u2, &q->x, &z1z1);
    f2_mul(&t, &p->z, &z1z1);
    f2_mul(&s2, &q->y, &t);
    f2_sub(&h, &u2, &p->x);
    f2_sub(&r, &s2, &p->y);
    if (f2_is_zero(&h)) {
        if (f2_is_zero(&r)) { bg2_dbl(o, p); return; }
        bg2_set_inf(o);
        return;
    }
    f2 hh, hhh, v, x3, y3, z3;
    f2_sqr(&hh, &h);
    f2_mul(&hhh, &h, &hh);
    f2_mul(&v, &p->x, &hh);
    f2_sqr(&x3, &r);
    f2_sub(&x3, &x3, &hhh);
    f2_sub(&x3, &x3, &v);
    f2_sub(&x3, &x3, &v);
    f2_sub(&t, &v, &x3);
    f2_mul(&y3, &r, &t);
    f2_mul(&t, &p->y, &hhh);
    f2_sub(&y3, &y3, &t);
    f2_mul(&z3, &p->z, &h);
    o->x = x3;
    o->y = y3;
    o->z = z3;
}

static int bg2a_on_curve(const bg2a *p) {
    f2 l, r, b2;
    f2_sqr(&l, &p->y);
    f2_sqr(&r, &p->x);
    f2_mul(&r, &r, &p->x);
    fb_from_limbs(&b2.c0, BLS_B4_MONT);
    fb_from_limbs(&b2.c1, BLS_B4_MONT);
    f2_add(&r, &r, &b2);
    return f2_eq(&l, &r);
}

static void bg2_to_affine_be(uint8_t out[192], const bg2j *p) {
    if (bg2_is_inf(p)) { memset(out, 0, 192); return; }
    f2 zi, zi2, zi3, xa, ya;
    fb c;
    f2_inv(&zi, &p->z);
    f2_sqr(&zi2, &zi);
    f2_mul(&zi3, &zi2, &zi);
    f2_mul(&xa, &p->x, &zi2);
    f2_mul(&ya, &p->y, &zi3);
    fb_from_mont(&c, &xa.c0); fb_to_be(out, &c);
    fb_from_mont(&c, &xa.c1); fb_to_be(out + 48, &c);
    fb_from_mont(&c, &ya.c0); fb_to_be(out + 96, &c);
    fb_from_mont(&c, &ya.c1); fb_to_be(out + 144, &c);
}

static int bg2_parse_be(bg2a *o, int *is_inf, const uint8_t in[192]) {
    fb v[4], m;
    fb_from_limbs(&m, BLSP_MOD);
    for (int k = 0; k < 4; k++) {
        fb_from_be(&v[k], in + 48 * k);
        if (fb_cmp(&v[k], &m) >= 0) return BORC_ERR_INPUT;
    }
    if (fb_is_zero(&v[0]) && fb_is_zero(&v[1]) && fb_is_zero(&v[2]) &&
        fb_is_zero(&v[3])) {
        *is_inf = 1;
        return BORC_OK;
    }
    fb_to_mont(&o->x.c0, &v[0]);
    fb_to_mont(&o->x.c1, &v[1]);
    fb_to_mont(&o->y.c0, &v[2]);
    fb_to_mont(&o->y.c1, &v[3]);
    *is_inf = 0;
    if (!bg2a_on_curve(o)) return BORC_ERR_POINT;
    return BORC_OK;
}

static void bg2_scalar_mul_be(bg2j *o, const bg2a *p, const uint8_t k[32]) {
    bg2j acc;
    bg2_set_inf(&acc);
    for (int i = 0; i < 256; i++) {
        bg2_dbl(&acc, &acc);
        if ((k[i / 8] >> (7 - (i % 8))) & 1) bg2_add_affine(&acc, &acc, p);
    }
    *o = acc;
}

static void bg2_full_add(bg2j *o, const bg2j *p, const bg2j *q) {
    if (bg2_is_inf(p)) { *o = *q; return; }
    if (bg2_is_inf(q)) { *o = *p; return; }
    f2 z1z1, z2z2, u1, u2, s1, s2, h, r, t;
    f2_sqr(&z1z1, &p->z);
    f2_sqr(&z2z2, &q->z);
    f2_mul(&u1, &p->x, &z2z2);
    f2_mul(&u2, &q->x, &z1z1);
    f2_mul(&t, &q->z, &z2z2);
    f2_mul(&s1, &p->y, &t);
    f2_mul(&t, &p->z, &z1z1);
    f2_mul(&s2, &q->y, &t);
    f2_sub(&h, &u2, &u1);
    f2_sub(&r, &s2, &s1);
    if (f2_is_zero(&h)) {
        if (f2_is_zero(&r)) { bg2_dbl(o, p); return; }
        bg2_set_inf(o);
        return;
    }
    f2 hh, hhh, v, x3, y3, z3;
    f2_sqr(&hh, &h);
    f2_mul(&hhh, &h, &hh);
    f2_mul(&v, &u1, &hh);
    f2_sqr(&x3, &r);
    f2_sub(&x3, &x3, &hhh);
    f2_sub(&x3, &x3, &v);
    f2_sub(&x3, &x3, &v);
    f2_sub(&t, &v, &x3);
    f2_mul(&y3, &r, &t);
    f2_mul(&t, &s1, &hhh);
    f2_sub(&y3, &y3, &t);
    f2_mul(&z3, &p->z, &q->z);
    f2_mul(&z3, &z3, &h);
    o->x = x3;
    o->y = y3;
    o->z = z3;
}

static int bg2_in_subgroup(const bg2a *p) {
    uint8_t rbe[32];
    for (int i = 0; i < 4; i++)
        for (int j = 0; j < 8; j++)
            rbe[(3 - i) * 8 + j] = (uint8_t)(BLSR_MOD[i] >> (56 - 8 * j));
    bg2j t;
    bg2_scalar_mul_be(&t, p, rbe);
    return bg2_is_inf(&t);
}

int oracle_bls_g2_add(const uint8_t p1[192], const uint8_t p2[192],
                      uint8_t out[192]) {
    bg2a a, b;
    int ia, ib, rc;
    if ((rc = bg2_parse_be(&a, &ia, p1))) return rc;
    if ((rc = bg2_parse_be(&b, &ib, p2))) return rc;
    bg2j acc;
    bg2_set_inf(&acc);
    if (!ia) bg2_add_affine(&acc, &acc, &a);
    if (!ib) bg2_add_affine(&acc, &acc, &b);
    bg2_to_affine_be(out, &acc);
    return BORC_OK;
}

int oracle_bls_g2_mul(const uint8_t point[192], const uint8_t scalar[32],
                      uint8_t out[192]) {
    bg2a p;
    int inf, rc;
    if ((rc = bg2_parse_be(&p, &inf, point))) return rc;
    if (inf) { memset(out, 0, 192); return BORC_OK; }
    bg2j r;
    bg2_scalar_mul_be(&r, &p, scalar);
    bg2_to_affine_be(out, &r);
    return BORC_OK;
}

/* Pippenger c=16, raw 256-bit scalars, per-point subgroup check
 * (bls_blst.rs g2_msm -> read_g2_subgroup) */
int oracle_bls_g2_msm(const uint8_t *points, const uint8_t *scalars, size_t n,
                      uint8_t out[192]) {
    bg2a *pts = malloc(n * sizeof(bg2a));
    uint8_t *inf = malloc(n);
    int err = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 16)
#endif
    for (size_t i = 0; i < n; i++) {
        int ii, rc;
        if ((rc = bg2_parse_be(&pts[i], &ii, points + 192 * i))) {
#ifdef _OPENMP
#pragma omp atomic write
#endif
            err = rc;
            continue;
        }
        inf[i] = (uint8_t)ii;
        if (!ii && !bg2_in_subgroup(&pts[i])) {
#ifdef _OPENMP
#pragma omp atomic write
#endif
            err = BORC_ERR_SUBGROUP;
        }
    }
    if (err) { free(pts); free(inf); return err; }
    bg2j wsum[BNWIN];
#ifdef _OPENMP
#pragma omp parallel
#endif
    {
        bg2j *buckets = malloc(BNBUCKET * sizeof(bg2j));
#ifdef _OPENMP
#pragma omp for schedule(dynamic)
#endif
        for (int w = 0; w < BNWIN; w++) {
            for (uint32_t b = 0; b < BNBUCKET; b++) bg2_set_inf(&buckets[b]);
            for (size_t i = 0; i < n; i++) {
                if (inf[i]) continue;
                const uint8_t *k = scalars + 32 * i;
                int bit = BC * w;
                uint32_t d = 0;
                for (int t = 0; t < BC; t++) {
                    int bb = bit + t;
                    d |= (uint32_t)((k[31 - bb / 8] >> (bb % 8)) & 1) << t;
                }
                if (d) bg2_add_affine(&buckets[d - 1], &buckets[d - 1], &pts[i]);
            }
            bg2j runj, sumj;
            bg2_set_inf(&runj);
            bg2_set_inf(&sumj);
            for (uint32_t b = BNBUCKET; b-- > 0;) {
                /* runj += buckets[b]; sumj += runj (full Jacobian add) */
                bg2j tmp;
                bg2_full_add(&tmp, &runj, &buckets[b]);
                runj = tmp;
                bg2_full_add(&tmp, &sumj, &runj);
                sumj = tmp;
            }
            wsum[w] = sumj;
        }
        free(buckets);
    }
    free(pts);
    free(inf);
    bg2j acc = wsum[BNWIN - 1];
    for (int w = BNWIN - 2; w >= 0; w--) {
        for (int t = 0; t < BC; t++) bg2_dbl(&acc, &acc);
        bg2j tmp;
        bg2_full_add(&tmp, &acc, &wsum[w]);
        acc = tmp;
    }
    bg2_to_affine_be(out, &acc);
    return BORC_OK;
}

int oracle_bls_g2_gen_points(uint64_t start, size_t n, uint8_t *out) {
    if (n == 0) return BORC_OK;
    bg2a gen;
    fb_from_limbs(&gen.x.c0, BLS_G2X0_MONT);
    fb_from_limbs(&gen.x.c1, BLS_G2X1_MONT);
    fb_from_limbs(&gen.y.c0, BLS_G2Y0_MONT);
    fb_from_limbs(&gen.y.c1, BLS_G2Y1_MONT);
    bg2j *acc = malloc(n * sizeof(bg2j));
    uint8_t k[32] = {0};
    uint64_t k0 = start + 1;
    for (int j = 0; j < 8; j++) k[31 - j] = (uint8_t)(k0 >> (8 * j));
    bg2_scalar_mul_be(&acc[0], &gen, k);
    for (size_t i = 1; i < n; i++) bg2_add_affine(&acc[i], &acc[i - 1], &gen);
    for (size_t i = 0; i < n; i++) bg2_to_affine_be(out + 192 * i, &acc[i]);
    free(acc);
    return BORC_OK;
}
