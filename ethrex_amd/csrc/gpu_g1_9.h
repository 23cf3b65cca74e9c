// ============================================================================
// BN254 G1 on the fe9 (9x29-bit) field core — XYZZ coordinates, Montgomery
// 2^261.  XYZZ (x = X/ZZ, y = Y/ZZZ with ZZ = z^2, ZZZ = z^3) gives the
// cheapest mixed add on this field core: 10 muls vs 11 for Jacobian
// (EFD madd-2008-s / add-2008-s / dbl-2008-s shapes, a = 0).
//
// Reference semantics unchanged (provider.rs:247-318; (0,0) identity);
// byte-level results identical.  Bound audit per gpu_field9.h contracts is
// annotated at each sub call site.  infinity <=> ZZ ≡ 0 (mod p).
// ============================================================================
#pragma once
#include "gpu_field9.h"

namespace em {

struct g1a9 {
    fe9 x, y;  // affine, Montgomery form, norm2p
};

struct g1j9 {               // name kept for kernel compatibility; XYZZ layout
    fe9 x, y, zz, zzz;
};

__device__ __forceinline__ g1j9 g1_inf9() {
    g1j9 p;
    p.x = fe9_load(bn254::FQ9_ONE);
    p.y = fe9_load(bn254::FQ9_ONE);
    p.zz = fe9_zero();
    p.zzz = fe9_zero();
    return p;
}

__device__ __forceinline__ bool g1_is_inf9(const g1j9 &p) {
    return fe9_is_zero_modp(p.zz);
}

// doubling (dbl-2008-s, a = 0)
__device__ __forceinline__ g1j9 g1_dbl9(const g1j9 &p) {
    if (g1_is_inf9(p)) return p;
    fe9 U = add9(p.y, p.y);                    // lazy <=2^30
    fe9 V = mont_mul9(U, U);                   // (2Y)^2
    fe9 W = mont_mul9(U, V);                   // (2Y)^3
    fe9 S = mont_mul9(p.x, V);
    fe9 A = mont_sqr9(p.x);
    fe9 M = add9_n(add9(A, A), A);             // 3X^2, norm2p
    g1j9 o;
    o.x = subm9(subm9(mont_sqr9(M), S), S);    // M^2 - 2S
    o.y = subm9(mont_mul9(M, subn9(S, o.x)), mont_mul9(W, p.y));
    o.zz = mont_mul9(V, p.zz);
    o.zzz = mont_mul9(W, p.zzz);
    return o;
}

// full XYZZ + XYZZ (add-2008-s)
__device__ __forceinline__ g1j9 g1_add9(const g1j9 &p, const g1j9 &q) {
    if (g1_is_inf9(p)) return q;
    if (g1_is_inf9(q)) return p;
    fe9 u1 = mont_mul9(p.x, q.zz);
    fe9 u2 = mont_mul9(q.x, p.zz);
    fe9 s1 = mont_mul9(p.y, q.zzz);
    fe9 s2 = mont_mul9(q.y, p.zzz);
    fe9 P = subm9(u2, u1);                     // b=u1 mul-out
    fe9 R = subm9(s2, s1);
    if (__builtin_expect(fe9_is_zero_modp(P), 0)) {
        if (fe9_is_zero_modp(R)) return g1_dbl9(p);
        return g1_inf9();
    }
    fe9 PP = mont_sqr9(P);
    fe9 PPP = mont_mul9(P, PP);
    fe9 Q = mont_mul9(u1, PP);
    g1j9 o;
    o.x = subm9(subm9(subm9(mont_sqr9(R), PPP), Q), Q);
    o.y = subm9(mont_mul9(R, subn9(Q, o.x)), mont_mul9(s1, PPP));
    o.zz = mont_mul9(mont_mul9(p.zz, q.zz), PP);
    o.zzz = mont_mul9(mont_mul9(p.zzz, q.zzz), PPP);
    return o;
}

// mixed add (madd-2008-s): q affine, not infinity
__device__ __forceinline__ g1j9 g1_add_affine9(const g1j9 &p, const g1a9 &q) {
    if (__builtin_expect(g1_is_inf9(p), 0)) {
        g1j9 o;
        o.x = q.x;
        o.y = q.y;
        o.zz = fe9_load(bn254::FQ9_ONE);
        o.zzz = fe9_load(bn254::FQ9_ONE);
        return o;
    }
    fe9 u2 = mont_mul9(q.x, p.zz);
    fe9 s2 = mont_mul9(q.y, p.zzz);
    fe9 P = subn9(u2, p.x);                    // b=X1 norm2p
    fe9 R = subn9(s2, p.y);
    if (__builtin_expect(fe9_is_zero_modp(P), 0)) {
        if (fe9_is_zero_modp(R)) return g1_dbl9(p);
        return g1_inf9();
    }
    fe9 PP = mont_sqr9(P);
    fe9 PPP = mont_mul9(P, PP);
    fe9 Q = mont_mul9(p.x, PP);
    g1j9 o;
    o.x = subm9(subm9(subm9(mont_sqr9(R), PPP), Q), Q);
    o.y = subm9(mont_mul9(R, subn9(Q, o.x)), mont_mul9(p.y, PPP));
    o.zz = mont_mul9(p.zz, PP);
    o.zzz = mont_mul9(p.zzz, PPP);
    return o;
}

// y^2 == x^3 + 3 (mod p), inputs norm2p
__device__ __forceinline__ bool g1a9_on_curve(const g1a9 &p) {
    fe9 l = mont_sqr9(p.y);
    fe9 r = mont_mul9(mont_sqr9(p.x), p.x);
    r = add9_n(r, fe9_load(bn254::FQ9_B3));
    return fe9_eq_modp(l, r);
}

__device__ __forceinline__ g1a9 g1_generator9() {
    g1a9 g;
    g.x = fe9_load(bn254::FQ9_GX);
    g.y = fe9_load(bn254::FQ9_GY);
    return g;
}

// scalar mul, k canonical 4x64, p affine non-infinity
__device__ __forceinline__ g1j9 g1_scalar_mul9(const g1a9 &p, const u64 k[4]) {
    g1j9 acc = g1_inf9();
    for (int i = 255; i >= 0; i--) {
        acc = g1_dbl9(acc);
        if ((k[i >> 6] >> (i & 63)) & 1) acc = g1_add_affine9(acc, p);
    }
    return acc;
}

// big-endian byte output helpers (canonical form)
__device__ __forceinline__ void fe9_to_be(uint8_t *b, const fe9 &canon) {
    u64 w[4];
    fe9_to_u64x4(w, canon);
    u64 *o = (u64 *)b;
    o[0] = __builtin_bswap64(w[3]);
    o[1] = __builtin_bswap64(w[2]);
    o[2] = __builtin_bswap64(w[1]);
    o[3] = __builtin_bswap64(w[0]);
}

__device__ __forceinline__ fe9 fe9_from_be(const uint8_t *b) {
    const u64 *w = (const u64 *)b;
    u64 v[4] = {__builtin_bswap64(w[3]), __builtin_bswap64(w[2]),
                __builtin_bswap64(w[1]), __builtin_bswap64(w[0])};
    return fe9_from_u64x4(v);
}

// XYZZ -> affine (x = X/ZZ, y = Y/ZZZ): one inversion + 3 muls
__device__ __forceinline__ g1a9 g1_to_affine9(const g1j9 &p) {
    fe9 t = mont_inv9(mont_mul9(p.zz, p.zzz));  // 1/(ZZ*ZZZ)
    g1a9 a;
    a.x = fe9_csub2p(mont_mul9(p.x, mont_mul9(t, p.zzz)));
    a.y = fe9_csub2p(mont_mul9(p.y, mont_mul9(t, p.zz)));
    return a;
}

// XYZZ -> affine 64-byte BE; infinity -> zeros
__device__ __forceinline__ void g1_to_affine_be9(uint8_t *out, const g1j9 &p) {
    if (g1_is_inf9(p)) {
        for (int i = 0; i < 8; i++) ((u64 *)out)[i] = 0;
        return;
    }
    g1a9 a = g1_to_affine9(p);
    fe9_to_be(out, from_mont9(a.x));
    fe9_to_be(out + 32, from_mont9(a.y));
}

// XYZZ -> Jacobian (X_j, Y_j, Z_j) with Z_j = ZZ*ZZZ (no inversion):
//   X_j = x*Z_j^2 = X*ZZ*ZZZ^2,  Y_j = y*Z_j^3 = Y*ZZ^3*ZZZ^2
__device__ __forceinline__ void g1_xyzz_to_jacobian9(fe9 &X, fe9 &Y, fe9 &Z,
                                                     const g1j9 &p) {
    fe9 zzz2 = mont_sqr9(p.zzz);
    fe9 zz2 = mont_sqr9(p.zz);
    X = mont_mul9(mont_mul9(p.x, p.zz), zzz2);
    Y = mont_mul9(mont_mul9(p.y, mont_mul9(zz2, p.zz)), zzz2);
    Z = mont_mul9(p.zz, p.zzz);
}

// Jacobian (X, Y, Z) -> XYZZ: ZZ = Z^2, ZZZ = Z^3; adjust X,Y? no —
// Jacobian x = X/Z^2 = X/ZZ, y = Y/Z^3 = Y/ZZZ: same numerators.
__device__ __forceinline__ g1j9 g1_jacobian_to_xyzz9(const fe9 &X, const fe9 &Y,
                                                     const fe9 &Z) {
    g1j9 p;
    p.x = X;
    p.y = Y;
    p.zz = mont_sqr9(Z);
    p.zzz = mont_mul9(p.zz, Z);
    return p;
}

}  // namespace em
