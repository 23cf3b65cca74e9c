// ============================================================================
// Device BN254 G1 ops — Jacobian coordinates over Fq, Montgomery form.
// Curve y^2 = x^3 + 3, generator G = (1,2)
// (reference semantics: crates/common/crypto/provider.rs:247-318;
//  identity encoded (0,0): crates/vm/levm/src/precompiles.rs:792-795).
// infinity <=> Z == 0.
// ============================================================================
#pragma once
#include "gpu_field.h"

namespace em {

using Fq = bn254::Fq;
using Fr = bn254::Fr;

struct g1a {
    fe4 x, y;  // affine, Montgomery form
};

struct g1j {
    fe4 x, y, z;  // Jacobian, Montgomery form; z==0 => infinity
};

__device__ __forceinline__ g1j g1_inf() {
    g1j p;
    p.x = fe_one_mont<Fq>();
    p.y = fe_one_mont<Fq>();
    p.z = fe4{{0, 0, 0, 0}};
    return p;
}

__device__ __forceinline__ bool g1_is_inf(const g1j &p) { return fe_is_zero(p.z); }

// doubling (a = 0): A=X^2 B=Y^2 C=B^2 D=2((X+B)^2-A-C) E=3A F=E^2
__device__ __forceinline__ g1j g1_dbl(const g1j &p) {
    if (g1_is_inf(p)) return p;
    fe4 A = mont_sqr<Fq>(p.x);
    fe4 B = mont_sqr<Fq>(p.y);
    fe4 C = mont_sqr<Fq>(B);
    fe4 t = mod_add<Fq>(p.x, B);
    t = mont_sqr<Fq>(t);
    t = mod_sub<Fq>(t, A);
    t = mod_sub<Fq>(t, C);
    fe4 D = mod_dbl<Fq>(t);
    fe4 E = mod_add<Fq>(mod_dbl<Fq>(A), A);
    fe4 F = mont_sqr<Fq>(E);
    g1j o;
    o.x = mod_sub<Fq>(mod_sub<Fq>(F, D), D);
    fe4 y3 = mont_mul<Fq>(E, mod_sub<Fq>(D, o.x));
    fe4 c8 = mod_dbl<Fq>(mod_dbl<Fq>(mod_dbl<Fq>(C)));
    o.y = mod_sub<Fq>(y3, c8);
    o.z = mod_dbl<Fq>(mont_mul<Fq>(p.y, p.z));
    return o;
}

// full Jacobian + Jacobian
__device__ __forceinline__ g1j g1_add(const g1j &p, const g1j &q) {
    if (g1_is_inf(p)) return q;
    if (g1_is_inf(q)) return p;
    fe4 z1z1 = mont_sqr<Fq>(p.z);
    fe4 z2z2 = mont_sqr<Fq>(q.z);
    fe4 u1 = mont_mul<Fq>(p.x, z2z2);
    fe4 u2 = mont_mul<Fq>(q.x, z1z1);
    fe4 s1 = mont_mul<Fq>(p.y, mont_mul<Fq>(q.z, z2z2));
    fe4 s2 = mont_mul<Fq>(q.y, mont_mul<Fq>(p.z, z1z1));
    fe4 h = mod_sub<Fq>(u2, u1);
    fe4 r = mod_sub<Fq>(s2, s1);
    if (fe_is_zero(h)) {
        if (fe_is_zero(r)) return g1_dbl(p);
        return g1_inf();
    }
    fe4 hh = mont_sqr<Fq>(h);
    fe4 hhh = mont_mul<Fq>(h, hh);
    fe4 v = mont_mul<Fq>(u1, hh);
    g1j o;
    o.x = mod_sub<Fq>(mod_sub<Fq>(mod_sub<Fq>(mont_sqr<Fq>(r), hhh), v), v);
    o.y = mod_sub<Fq>(mont_mul<Fq>(r, mod_sub<Fq>(v, o.x)),
                      mont_mul<Fq>(s1, hhh));
    o.z = mont_mul<Fq>(mont_mul<Fq>(p.z, q.z), h);
    return o;
}

// mixed add: q affine (implicit z=1), q must not be infinity
__device__ __forceinline__ g1j g1_add_affine(const g1j &p, const g1a &q) {
    if (__builtin_expect(g1_is_inf(p), 0)) {
        g1j o;
        o.x = q.x;
        o.y = q.y;
        o.z = fe_one_mont<Fq>();
        return o;
    }
    fe4 z1z1 = mont_sqr<Fq>(p.z);
    fe4 u2 = mont_mul<Fq>(q.x, z1z1);
    fe4 s2 = mont_mul<Fq>(q.y, mont_mul<Fq>(p.z, z1z1));
    fe4 h = mod_sub<Fq>(u2, p.x);
    fe4 r = mod_sub<Fq>(s2, p.y);
    if (__builtin_expect(fe_is_zero(h), 0)) {
        if (fe_is_zero(r)) return g1_dbl(p);
        return g1_inf();
    }
    fe4 hh = mont_sqr<Fq>(h);
    fe4 hhh = mont_mul<Fq>(h, hh);
    fe4 v = mont_mul<Fq>(p.x, hh);
    g1j o;
    o.x = mod_sub<Fq>(mod_sub<Fq>(mod_sub<Fq>(mont_sqr<Fq>(r), hhh), v), v);
    o.y = mod_sub<Fq>(mont_mul<Fq>(r, mod_sub<Fq>(v, o.x)),
                      mont_mul<Fq>(p.y, hhh));
    o.z = mont_mul<Fq>(p.z, h);
    return o;
}

// y^2 == x^3 + 3 for affine Montgomery point
__device__ __forceinline__ bool g1a_on_curve(const g1a &p) {
    fe4 l = mont_sqr<Fq>(p.y);
    fe4 r = mont_mul<Fq>(mont_sqr<Fq>(p.x), p.x);
    r = mod_add<Fq>(r, fe4{{bn254::FQ_B3_MONT[0], bn254::FQ_B3_MONT[1],
                            bn254::FQ_B3_MONT[2], bn254::FQ_B3_MONT[3]}});
    return fe_eq(l, r);
}

__device__ __forceinline__ g1a g1_generator() {
    g1a g;
    g.x = fe4{{bn254::FQ_GX_MONT[0], bn254::FQ_GX_MONT[1], bn254::FQ_GX_MONT[2],
               bn254::FQ_GX_MONT[3]}};
    g.y = fe4{{bn254::FQ_GY_MONT[0], bn254::FQ_GY_MONT[1], bn254::FQ_GY_MONT[2],
               bn254::FQ_GY_MONT[3]}};
    return g;
}

// scalar mul, k canonical 4x64, p affine non-infinity
__device__ __forceinline__ g1j g1_scalar_mul(const g1a &p, const fe4 &k) {
    g1j acc = g1_inf();
    for (int i = 255; i >= 0; i--) {
        acc = g1_dbl(acc);
        if ((k.v[i >> 6] >> (i & 63)) & 1) acc = g1_add_affine(acc, p);
    }
    return acc;
}

// Jacobian -> affine big-endian 64 bytes; infinity -> zeros
__device__ __forceinline__ void g1_to_affine_be(uint8_t *out, const g1j &p) {
    if (g1_is_inf(p)) {
        for (int i = 0; i < 8; i++) ((u64 *)out)[i] = 0;
        return;
    }
    fe4 zi = mont_inv<Fq>(p.z);
    fe4 zi2 = mont_sqr<Fq>(zi);
    fe4 zi3 = mont_mul<Fq>(zi2, zi);
    fe_to_be(out, from_mont<Fq>(mont_mul<Fq>(p.x, zi2)));
    fe_to_be(out + 32, from_mont<Fq>(mont_mul<Fq>(p.y, zi3)));
}

}  // namespace em
