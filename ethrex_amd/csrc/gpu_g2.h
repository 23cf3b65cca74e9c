// ============================================================================
// BLS12-381 G2 on gfx950: Fp2 = Fp[u]/(u^2 + 1) over the 14x29-limb base
// field, curve y^2 = x^3 + 4(1+u).  EIP-2537 / blst byte semantics
// (crates/common/crypto/bls_blst.rs:226-255,303-321): 192-byte points
// x.c0||x.c1||y.c0||y.c1, canonical 48-B BE coords, (0,0,0,0) identity.
//
// The Pippenger machinery in msm_kernels.h / api.hip is curve-templated on
// the point structs g1aT<C>/g1jT<C> and the group-op overload set; G2 plugs
// in by SPECIALIZING those for the BlsG2 tag with fp2 coordinates — the
// digit/sort/offset/schedule pipeline is shared untouched.
//
// Bound discipline (gpu_field9.h contracts, L=14: mul inputs must be
// limb-normalized): every fp2 component stays norm2p; Karatsuba mul feeds
// add9_n outputs (norm2p) into mont_mul9.
// ============================================================================
#pragma once
#include "gpu_g1_9.h"

namespace em {

struct fp2 {
    fe14 c0, c1;
};

using F2B = FpB14T;

__device__ __host__ __forceinline__ fp2 fp2_zero() {
    return {fe9z<14>(), fe9z<14>()};
}
__device__ __host__ __forceinline__ fp2 fp2_one() {
    return {fe9_load<14>(F2B::ONE), fe9z<14>()};
}
__device__ __host__ __forceinline__ fp2 fp2_add_n(const fp2 &a, const fp2 &b) {
    return {add9_n<F2B>(a.c0, b.c0), add9_n<F2B>(a.c1, b.c1)};
}
// a - b, b a MUL-OUTPUT pair (< 1.5p per component)
__device__ __host__ __forceinline__ fp2 fp2_subm(const fp2 &a, const fp2 &b) {
    return {subm9<F2B>(a.c0, b.c0), subm9<F2B>(a.c1, b.c1)};
}
// a - b, b norm2p
__device__ __host__ __forceinline__ fp2 fp2_subn(const fp2 &a, const fp2 &b) {
    return {subn9<F2B>(a.c0, b.c0), subn9<F2B>(a.c1, b.c1)};
}
__device__ __host__ __forceinline__ bool fp2_is_zero_modp(const fp2 &a) {
    return fe9_is_zero_modp<F2B>(a.c0) && fe9_is_zero_modp<F2B>(a.c1);
}
__device__ __host__ __forceinline__ bool fp2_eq_modp(const fp2 &a,
                                                     const fp2 &b) {
    return fe9_eq_raw<14>(fe9_csubp<F2B>(a.c0), fe9_csubp<F2B>(b.c0)) &&
           fe9_eq_raw<14>(fe9_csubp<F2B>(a.c1), fe9_csubp<F2B>(b.c1));
}
// Karatsuba: 3 base muls; (a0+a1)(b0+b1) - a0b0 - a1b1 = a0b1 + a1b0
__device__ __host__ __forceinline__ fp2 fp2_mul(const fp2 &a, const fp2 &b) {
    fe14 m0 = mont_mul9<F2B>(a.c0, b.c0);
    fe14 m1 = mont_mul9<F2B>(a.c1, b.c1);
    fe14 m2 = mont_mul9<F2B>(add9_n<F2B>(a.c0, a.c1), add9_n<F2B>(b.c0, b.c1));
    fp2 r;
    r.c0 = subm9<F2B>(m0, m1);
    r.c1 = subm9<F2B>(subm9<F2B>(m2, m0), m1);
    return r;
}
// (a0+a1)(a0-a1), 2 a0 a1: 2 base muls
__device__ __host__ __forceinline__ fp2 fp2_sqr(const fp2 &a) {
    fe14 u = add9_n<F2B>(a.c0, a.c1);
    fe14 v = subn9<F2B>(a.c0, a.c1);
    fe14 w = mont_mul9<F2B>(a.c0, a.c1);
    fp2 r;
    r.c0 = mont_mul9<F2B>(u, v);
    r.c1 = add9_n<F2B>(w, w);
    return r;
}
// 1/(a0 + a1 u) = (a0 - a1 u) / (a0^2 + a1^2)
__device__ __host__ __forceinline__ fp2 fp2_inv(const fp2 &a) {
    fe14 n = add9_n<F2B>(mont_sqr9<F2B>(a.c0), mont_sqr9<F2B>(a.c1));
    fe14 t = mont_inv9<F2B>(n);
    fp2 r;
    r.c0 = mont_mul9<F2B>(a.c0, t);
    r.c1 = neg9<F2B>(fe9_csub2p<F2B>(mont_mul9<F2B>(a.c1, t)));
    return r;
}

// curve tag; F = the base field trait (scalar machinery uses F::W64)
struct BlsG2 {
    using F = FpB14T;
};

template <>
struct g1aT<BlsG2> {
    fp2 x, y;
};
template <>
struct g1jT<BlsG2> {
    fp2 x, y, zz, zzz;  // XYZZ over Fp2
};

template <>
__device__ __host__ __forceinline__ g1jT<BlsG2> g1_inf9<BlsG2>() {
    g1jT<BlsG2> p;
    p.x = fp2_one();
    p.y = fp2_one();
    p.zz = fp2_zero();
    p.zzz = fp2_zero();
    return p;
}

template <>
__device__ __host__ __forceinline__ bool g1_is_inf9<BlsG2>(const g1jT<BlsG2> &p) {
    return fp2_is_zero_modp(p.zz);
}

// dbl-2008-s over fp2 (a = 0)
template <>
__device__ __host__ __forceinline__ g1jT<BlsG2> g1_dbl9<BlsG2>(const g1jT<BlsG2> &p) {
    if (g1_is_inf9<BlsG2>(p)) return p;
    fp2 U = fp2_add_n(p.y, p.y);
    fp2 V = fp2_sqr(U);
    fp2 W = fp2_mul(U, V);
    fp2 S = fp2_mul(p.x, V);
    fp2 A = fp2_sqr(p.x);
    fp2 M = fp2_add_n(fp2_add_n(A, A), A);
    g1jT<BlsG2> o;
    o.x = fp2_subm(fp2_subm(fp2_sqr(M), S), S);
    o.y = fp2_subm(fp2_mul(M, fp2_subn(S, o.x)), fp2_mul(W, p.y));
    o.zz = fp2_mul(V, p.zz);
    o.zzz = fp2_mul(W, p.zzz);
    return o;
}

// add-2008-s over fp2
template <>
__device__ __host__ __forceinline__ g1jT<BlsG2> g1_add9<BlsG2>(const g1jT<BlsG2> &p,
                                                      const g1jT<BlsG2> &q) {
    if (g1_is_inf9<BlsG2>(p)) return q;
    if (g1_is_inf9<BlsG2>(q)) return p;
    fp2 u1 = fp2_mul(p.x, q.zz);
    fp2 u2 = fp2_mul(q.x, p.zz);
    fp2 s1 = fp2_mul(p.y, q.zzz);
    fp2 s2 = fp2_mul(q.y, p.zzz);
    fp2 P = fp2_subm(u2, u1);
    fp2 R = fp2_subm(s2, s1);
    if (__builtin_expect(fp2_is_zero_modp(P), 0)) {
        if (fp2_is_zero_modp(R)) return g1_dbl9<BlsG2>(p);
        return g1_inf9<BlsG2>();
    }
    fp2 PP = fp2_sqr(P);
    fp2 PPP = fp2_mul(P, PP);
    fp2 Q = fp2_mul(u1, PP);
    g1jT<BlsG2> o;
    o.x = fp2_subm(fp2_subm(fp2_subm(fp2_sqr(R), PPP), Q), Q);
    o.y = fp2_subm(fp2_mul(R, fp2_subn(Q, o.x)), fp2_mul(s1, PPP));
    o.zz = fp2_mul(fp2_mul(p.zz, q.zz), PP);
    o.zzz = fp2_mul(fp2_mul(p.zzz, q.zzz), PPP);
    return o;
}

// madd-2008-s over fp2
template <>
__device__ __host__ __forceinline__ g1jT<BlsG2> g1_add_affine9<BlsG2>(
    const g1jT<BlsG2> &p, const g1aT<BlsG2> &q) {
    if (__builtin_expect(g1_is_inf9<BlsG2>(p), 0)) {
        g1jT<BlsG2> o;
        o.x = q.x;
        o.y = q.y;
        o.zz = fp2_one();
        o.zzz = fp2_one();
        return o;
    }
    fp2 u2 = fp2_mul(q.x, p.zz);
    fp2 s2 = fp2_mul(q.y, p.zzz);
    fp2 P = fp2_subn(u2, p.x);
    fp2 R = fp2_subn(s2, p.y);
    if (__builtin_expect(fp2_is_zero_modp(P), 0)) {
        if (fp2_is_zero_modp(R)) return g1_dbl9<BlsG2>(p);
        return g1_inf9<BlsG2>();
    }
    fp2 PP = fp2_sqr(P);
    fp2 PPP = fp2_mul(P, PP);
    fp2 Q = fp2_mul(p.x, PP);
    g1jT<BlsG2> o;
    o.x = fp2_subm(fp2_subm(fp2_subm(fp2_sqr(R), PPP), Q), Q);
    o.y = fp2_subm(fp2_mul(R, fp2_subn(Q, o.x)), fp2_mul(p.y, PPP));
    o.zz = fp2_mul(p.zz, PP);
    o.zzz = fp2_mul(p.zzz, PPP);
    return o;
}

// y^2 == x^3 + 4(1+u)
template <>
__device__ __forceinline__ bool g1a9_on_curve<BlsG2>(const g1aT<BlsG2> &p) {
    fp2 l = fp2_sqr(p.y);
    fp2 r = fp2_mul(fp2_sqr(p.x), p.x);
    fp2 b2{fe9_load<14>(bn254::FPB_B4), fe9_load<14>(bn254::FPB_B4)};
    r = fp2_add_n(r, b2);
    return fp2_eq_modp(l, r);
}

template <>
__device__ __forceinline__ g1aT<BlsG2> g1_generator9<BlsG2>() {
    g1aT<BlsG2> g;
    g.x = {fe9_load<14>(bn254::FPB_G2X0), fe9_load<14>(bn254::FPB_G2X1)};
    g.y = {fe9_load<14>(bn254::FPB_G2Y0), fe9_load<14>(bn254::FPB_G2Y1)};
    return g;
}

// XYZZ -> affine (device; one fp2 inversion)
template <>
__device__ __host__ __forceinline__ g1aT<BlsG2> g1_to_affine9<BlsG2>(
    const g1jT<BlsG2> &p) {
    fp2 t = fp2_inv(fp2_mul(p.zz, p.zzz));
    g1aT<BlsG2> a;
    a.x = fp2_mul(p.x, fp2_mul(t, p.zzz));
    a.y = fp2_mul(p.y, fp2_mul(t, p.zz));
    return a;
}

// fp2 canonical byte IO (c0 || c1, 48-B BE each)
__device__ __host__ __forceinline__ void fp2_to_be(uint8_t *b, const fp2 &m) {
    feT_to_be<F2B>(b, fe9_csubp<F2B>(from_mont9<F2B>(m.c0)));
    feT_to_be<F2B>(b + 48, fe9_csubp<F2B>(from_mont9<F2B>(m.c1)));
}

// XYZZ -> affine BE bytes (192 B); infinity -> zeros
template <>
__device__ __host__ __forceinline__ void g1_to_affine_be9<BlsG2>(
    uint8_t *out, const g1jT<BlsG2> &p) {
    if (g1_is_inf9<BlsG2>(p)) {
        for (int i = 0; i < 24; i++) ((u64 *)out)[i] = 0;
        return;
    }
    g1aT<BlsG2> a = g1_to_affine9<BlsG2>(p);
    fp2_to_be(out, a.x);
    fp2_to_be(out + 96, a.y);
}

template <>
__device__ __host__ __forceinline__ void g1_neg_y9<BlsG2>(g1jT<BlsG2> &p) {
    p.y.c0 = neg9<F2B>(p.y.c0);
    p.y.c1 = neg9<F2B>(p.y.c1);
}

template <>
struct pt_bytes<BlsG2> {
    static constexpr int NB = 48;
    static constexpr int AFF = 192;  // x.c0||x.c1||y.c0||y.c1
    static constexpr int JAC = 288;  // X||Y||Z, each c0||c1
};

// Jacobian wire IO over Fp2 (288 B: X.c0||X.c1||Y.c0||Y.c1||Z.c0||Z.c1)
template <>
__device__ __host__ __forceinline__ void g1_jac_be9<BlsG2>(uint8_t *out,
                                                  const g1jT<BlsG2> &p) {
    if (g1_is_inf9<BlsG2>(p)) {
        for (int j = 0; j < 36; j++) ((u64 *)out)[j] = 0;
        return;
    }
    fp2 zzz2 = fp2_sqr(p.zzz);
    fp2 X = fp2_mul(fp2_mul(p.x, p.zz), zzz2);
    fp2 Y = fp2_mul(fp2_mul(p.y, fp2_mul(fp2_sqr(p.zz), p.zz)), zzz2);
    fp2 Z = fp2_mul(p.zz, p.zzz);
    fp2_to_be(out, X);
    fp2_to_be(out + 96, Y);
    fp2_to_be(out + 192, Z);
}

__device__ __host__ __forceinline__ fp2 fp2_from_be_mont(const uint8_t *b) {
    fp2 r;
    r.c0 = to_mont9<F2B>(feT_from_be<F2B>(b));
    r.c1 = to_mont9<F2B>(feT_from_be<F2B>(b + 48));
    return r;
}

template <>
__device__ __host__ __forceinline__ bool g1_jac_from_be9<BlsG2>(g1jT<BlsG2> &o,
                                                       const uint8_t *in) {
    fp2 X = fp2_from_be_mont(in);
    fp2 Y = fp2_from_be_mont(in + 96);
    fp2 Z = fp2_from_be_mont(in + 192);
    if (fp2_is_zero_modp(Z)) return false;
    o.x = X;
    o.y = Y;
    o.zz = fp2_sqr(Z);
    o.zzz = fp2_mul(o.zz, Z);
    return true;
}

}  // namespace em
