// ============================================================================
// BN254 G1 on the fe9 (9x29-bit) field core — Jacobian, Montgomery 2^261.
// Same reference semantics as before (provider.rs:247-318; (0,0) identity):
// the byte-level results are identical, only the internal representation
// changed.  Every sub/add call site is annotated with the bound audit from
// gpu_field9.h's contracts.
// ============================================================================
#pragma once
#include "gpu_field9.h"

namespace em {

struct g1a9 {
    fe9 x, y;  // affine, Montgomery form, norm2p
};

struct g1j9 {
    fe9 x, y, z;  // Jacobian; z == 0 (exact) => infinity
};

__device__ __forceinline__ g1j9 g1_inf9() {
    g1j9 p;
    p.x = fe9_load(bn254::FQ9_ONE);
    p.y = fe9_load(bn254::FQ9_ONE);
    p.z = fe9_zero();
    return p;
}

__device__ __forceinline__ bool g1_is_inf9(const g1j9 &p) {
    return fe9_is_zero_modp(p.z);
}

// doubling (a = 0)
__device__ __forceinline__ g1j9 g1_dbl9(const g1j9 &p) {
    if (g1_is_inf9(p)) return p;
    fe9 A = mont_sqr9(p.x);                    // <1.01p
    fe9 B = mont_sqr9(p.y);
    fe9 C = mont_sqr9(B);
    fe9 t = add9(p.x, B);                      // lazy <=2^30 limbs, <4p
    fe9 t2 = mont_mul9(t, t);                  // mul input ok
    t2 = subm9(subm9(t2, A), C);               // b mul-outs -> norm2p
    fe9 D = add9_n(t2, t2);                    // norm2p
    fe9 E = add9_n(add9(A, A), A);             // 3A, norm2p
    fe9 F = mont_sqr9(E);
    g1j9 o;
    o.x = subn9(subn9(F, D), D);               // b=D norm2p
    fe9 c8 = add9_n(C, C);
    c8 = add9_n(c8, c8);
    c8 = add9_n(c8, c8);                       // 8C norm2p
    o.y = subm9(mont_mul9(E, subn9(D, o.x)), c8);
    o.z = mont_mul9(add9(p.y, p.y), p.z);      // 2YZ (lazy add ok as mul input)
    return o;
}

// full Jacobian + Jacobian
__device__ __forceinline__ g1j9 g1_add9(const g1j9 &p, const g1j9 &q) {
    if (g1_is_inf9(p)) return q;
    if (g1_is_inf9(q)) return p;
    fe9 z1z1 = mont_sqr9(p.z);
    fe9 z2z2 = mont_sqr9(q.z);
    fe9 u1 = mont_mul9(p.x, z2z2);
    fe9 u2 = mont_mul9(q.x, z1z1);
    fe9 s1 = mont_mul9(p.y, mont_mul9(q.z, z2z2));
    fe9 s2 = mont_mul9(q.y, mont_mul9(p.z, z1z1));
    fe9 h = subm9(u2, u1);                     // b=u1 mul-out -> norm2p
    fe9 r = subm9(s2, s1);
    if (__builtin_expect(fe9_is_zero_modp(h), 0)) {
        if (fe9_is_zero_modp(r)) return g1_dbl9(p);
        return g1_inf9();
    }
    fe9 hh = mont_sqr9(h);
    fe9 hhh = mont_mul9(h, hh);
    fe9 v = mont_mul9(u1, hh);
    g1j9 o;
    o.x = subm9(subm9(subm9(mont_sqr9(r), hhh), v), v);
    o.y = subm9(mont_mul9(r, subn9(v, o.x)), mont_mul9(s1, hhh));
    o.z = mont_mul9(mont_mul9(p.z, q.z), h);
    return o;
}

// mixed add: q affine (z=1 implicit, norm2p coords), q not infinity
__device__ __forceinline__ g1j9 g1_add_affine9(const g1j9 &p, const g1a9 &q) {
    if (__builtin_expect(g1_is_inf9(p), 0)) {
        g1j9 o;
        o.x = q.x;
        o.y = q.y;
        o.z = fe9_load(bn254::FQ9_ONE);
        return o;
    }
    fe9 z1z1 = mont_sqr9(p.z);
    fe9 u2 = mont_mul9(q.x, z1z1);
    fe9 s2 = mont_mul9(q.y, mont_mul9(p.z, z1z1));
    fe9 h = subn9(u2, p.x);                    // b=X1 norm2p
    fe9 r = subn9(s2, p.y);
    if (__builtin_expect(fe9_is_zero_modp(h), 0)) {
        if (fe9_is_zero_modp(r)) return g1_dbl9(p);
        return g1_inf9();
    }
    fe9 hh = mont_sqr9(h);
    fe9 hhh = mont_mul9(h, hh);
    fe9 v = mont_mul9(p.x, hh);
    g1j9 o;
    o.x = subm9(subm9(subm9(mont_sqr9(r), hhh), v), v);
    o.y = subm9(mont_mul9(r, subn9(v, o.x)), mont_mul9(p.y, hhh));
    o.z = mont_mul9(p.z, h);
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

// Jacobian -> affine 64-byte BE; infinity -> zeros
__device__ __forceinline__ void g1_to_affine_be9(uint8_t *out, const g1j9 &p) {
    if (g1_is_inf9(p)) {
        for (int i = 0; i < 8; i++) ((u64 *)out)[i] = 0;
        return;
    }
    fe9 zi = mont_inv9(p.z);
    fe9 zi2 = mont_sqr9(zi);
    fe9 zi3 = mont_mul9(zi2, zi);
    fe9_to_be(out, from_mont9(mont_mul9(p.x, zi2)));
    fe9_to_be(out + 32, from_mont9(mont_mul9(p.y, zi3)));
}

}  // namespace em
