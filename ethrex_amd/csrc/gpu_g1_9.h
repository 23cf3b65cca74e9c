// ============================================================================
// Short-Weierstrass G1 (y^2 = x^3 + b) on the 29-bit-limb field core —
// XYZZ coordinates (x = X/ZZ, y = Y/ZZZ), cheapest mixed add here:
// 10 muls (EFD madd-2008-s / add-2008-s / dbl-2008-s, a = 0).
//
// Curve instances:
//   Bn254G1 (Fq9T,  b=3, G=(1,2))          — provider.rs:247-318 semantics
//   BlsG1   (FpB14T, b=4, standard G)      — bls_blst.rs EIP-2537 semantics
// Byte-level results identical to the respective references; bound audit per
// gpu_field9.h contracts annotated at each sub call site.
// infinity <=> ZZ ≡ 0 (mod p).
//
// NOTE (L=14 column budget): lazy add9 results must NOT feed muls on the
// BLS field — the doubling formula below therefore uses add9_n for U.
// ============================================================================
#pragma once
#include "gpu_field9.h"

namespace em {

struct Bn254G1 {
    using F = Fq9T;
    static constexpr const u32 (&CB)[9] = bn254::FQ9_B3;   // b in Montgomery
    static constexpr const u32 (&GX)[9] = bn254::FQ9_GX;
    static constexpr const u32 (&GY)[9] = bn254::FQ9_GY;
};
struct BlsG1 {
    using F = FpB14T;
    static constexpr const u32 (&CB)[14] = bn254::FPB_B4;
    static constexpr const u32 (&GX)[14] = bn254::FPB_GX;
    static constexpr const u32 (&GY)[14] = bn254::FPB_GY;
};

template <typename C>
struct g1aT {
    feL<C::F::L> x, y;  // affine, Montgomery form, norm2p
};

template <typename C>
struct g1jT {
    feL<C::F::L> x, y, zz, zzz;  // XYZZ
};

using g1a9 = g1aT<Bn254G1>;
using g1j9 = g1jT<Bn254G1>;
using g1aB = g1aT<BlsG1>;
using g1jB = g1jT<BlsG1>;

template <typename C = Bn254G1>
__device__ __host__ __forceinline__ g1jT<C> g1_inf9() {
    g1jT<C> p;
    p.x = fe9_load<C::F::L>(C::F::ONE);
    p.y = fe9_load<C::F::L>(C::F::ONE);
    p.zz = fe9z<C::F::L>();
    p.zzz = fe9z<C::F::L>();
    return p;
}

template <typename C>
__device__ __host__ __forceinline__ bool g1_is_inf9(const g1jT<C> &p) {
    return fe9_is_zero_modp<typename C::F>(p.zz);
}

// doubling (dbl-2008-s, a = 0)
template <typename C>
__device__ __host__ __forceinline__ g1jT<C> g1_dbl9(const g1jT<C> &p) {
    using F = typename C::F;
    if (g1_is_inf9(p)) return p;
    feL<F::L> U = add9_n<F>(p.y, p.y);         // 2Y, norm2p (L=14 mul rule)
    feL<F::L> V = mont_mul9<F>(U, U);
    feL<F::L> W = mont_mul9<F>(U, V);
    feL<F::L> S = mont_mul9<F>(p.x, V);
    feL<F::L> A = mont_sqr9<F>(p.x);
    feL<F::L> M = add9_n<F>(add9_n<F>(A, A), A);  // 3X^2, norm2p
    g1jT<C> o;
    o.x = subm9<F>(subm9<F>(mont_sqr9<F>(M), S), S);
    o.y = subm9<F>(mont_mul9<F>(M, subn9<F>(S, o.x)), mont_mul9<F>(W, p.y));
    o.zz = mont_mul9<F>(V, p.zz);
    o.zzz = mont_mul9<F>(W, p.zzz);
    return o;
}

// full XYZZ + XYZZ (add-2008-s)
template <typename C>
__device__ __host__ __forceinline__ g1jT<C> g1_add9(const g1jT<C> &p, const g1jT<C> &q) {
    using F = typename C::F;
    if (g1_is_inf9(p)) return q;
    if (g1_is_inf9(q)) return p;
    feL<F::L> u1 = mont_mul9<F>(p.x, q.zz);
    feL<F::L> u2 = mont_mul9<F>(q.x, p.zz);
    feL<F::L> s1 = mont_mul9<F>(p.y, q.zzz);
    feL<F::L> s2 = mont_mul9<F>(q.y, p.zzz);
    feL<F::L> P = subm9<F>(u2, u1);            // b=u1 mul-out
    feL<F::L> R = subm9<F>(s2, s1);
    if (__builtin_expect(fe9_is_zero_modp<F>(P), 0)) {
        if (fe9_is_zero_modp<F>(R)) return g1_dbl9(p);
        return g1_inf9<C>();
    }
    feL<F::L> PP = mont_sqr9<F>(P);
    feL<F::L> PPP = mont_mul9<F>(P, PP);
    feL<F::L> Q = mont_mul9<F>(u1, PP);
    g1jT<C> o;
    o.x = subm9<F>(subm9<F>(subm9<F>(mont_sqr9<F>(R), PPP), Q), Q);
    o.y = subm9<F>(mont_mul9<F>(R, subn9<F>(Q, o.x)), mont_mul9<F>(s1, PPP));
    o.zz = mont_mul9<F>(mont_mul9<F>(p.zz, q.zz), PP);
    o.zzz = mont_mul9<F>(mont_mul9<F>(p.zzz, q.zzz), PPP);
    return o;
}

// mixed add (madd-2008-s): q affine, not infinity
template <typename C>
__device__ __host__ __forceinline__ g1jT<C> g1_add_affine9(const g1jT<C> &p,
                                                  const g1aT<C> &q) {
    using F = typename C::F;
    if (__builtin_expect(g1_is_inf9(p), 0)) {
        g1jT<C> o;
        o.x = q.x;
        o.y = q.y;
        o.zz = fe9_load<F::L>(F::ONE);
        o.zzz = fe9_load<F::L>(F::ONE);
        return o;
    }
    feL<F::L> u2 = mont_mul9<F>(q.x, p.zz);
    feL<F::L> s2 = mont_mul9<F>(q.y, p.zzz);
    feL<F::L> P = subn9<F>(u2, p.x);           // b=X1 norm2p
    feL<F::L> R = subn9<F>(s2, p.y);
    if (__builtin_expect(fe9_is_zero_modp<F>(P), 0)) {
        if (fe9_is_zero_modp<F>(R)) return g1_dbl9(p);
        return g1_inf9<C>();
    }
    feL<F::L> PP = mont_sqr9<F>(P);
    feL<F::L> PPP = mont_mul9<F>(P, PP);
    feL<F::L> Q = mont_mul9<F>(p.x, PP);
    g1jT<C> o;
    o.x = subm9<F>(subm9<F>(subm9<F>(mont_sqr9<F>(R), PPP), Q), Q);
    o.y = subm9<F>(mont_mul9<F>(R, subn9<F>(Q, o.x)), mont_mul9<F>(p.y, PPP));
    o.zz = mont_mul9<F>(p.zz, PP);
    o.zzz = mont_mul9<F>(p.zzz, PPP);
    return o;
}

// y^2 == x^3 + b (mod p), inputs norm2p
template <typename C>
__device__ __forceinline__ bool g1a9_on_curve(const g1aT<C> &p) {
    using F = typename C::F;
    feL<F::L> l = mont_sqr9<F>(p.y);
    feL<F::L> r = mont_mul9<F>(mont_sqr9<F>(p.x), p.x);
    r = add9_n<F>(r, fe9_load<F::L>(C::CB));
    return fe9_eq_modp<F>(l, r);
}

template <typename C = Bn254G1>
__device__ __forceinline__ g1aT<C> g1_generator9() {
    g1aT<C> g;
    g.x = fe9_load<C::F::L>(C::GX);
    g.y = fe9_load<C::F::L>(C::GY);
    return g;
}

// scalar mul, k canonical u64 words (nwords*64 bits scanned), p affine
template <typename C>
__device__ __forceinline__ g1jT<C> g1_scalar_mul9(const g1aT<C> &p,
                                                  const u64 *k, int nwords = 4) {
    g1jT<C> acc = g1_inf9<C>();
    for (int i = 64 * nwords - 1; i >= 0; i--) {
        acc = g1_dbl9(acc);
        if ((k[i >> 6] >> (i & 63)) & 1) acc = g1_add_affine9(acc, p);
    }
    return acc;
}

// big-endian byte IO of canonical field elements (T::W64 * 8 bytes)
template <typename T>
__device__ __host__ __forceinline__ void feT_to_be(uint8_t *b, const feL<T::L> &canon) {
    u64 w[T::W64];
    fe9_to_u64<T>(w, canon);
    u64 *o = (u64 *)b;
#pragma unroll
    for (int i = 0; i < T::W64; i++)
        o[i] = __builtin_bswap64(w[T::W64 - 1 - i]);
}

template <typename T>
__device__ __host__ __forceinline__ feL<T::L> feT_from_be(const uint8_t *b) {
    const u64 *w = (const u64 *)b;
    u64 v[T::W64];
#pragma unroll
    for (int i = 0; i < T::W64; i++)
        v[i] = __builtin_bswap64(w[T::W64 - 1 - i]);
    return fe9_from_u64<T>(v);
}

// legacy bn254 names
__device__ __host__ __forceinline__ void fe9_to_be(uint8_t *b, const fe9 &canon) {
    feT_to_be<Fq9T>(b, canon);
}
__device__ __host__ __forceinline__ fe9 fe9_from_be(const uint8_t *b) {
    return feT_from_be<Fq9T>(b);
}

// XYZZ -> affine (x = X/ZZ, y = Y/ZZZ): one inversion + 3 muls
template <typename C>
__device__ __host__ __forceinline__ g1aT<C> g1_to_affine9(const g1jT<C> &p) {
    using F = typename C::F;
    feL<F::L> t = mont_inv9<F>(mont_mul9<F>(p.zz, p.zzz));
    g1aT<C> a;
    a.x = fe9_csub2p<F>(mont_mul9<F>(p.x, mont_mul9<F>(t, p.zzz)));
    a.y = fe9_csub2p<F>(mont_mul9<F>(p.y, mont_mul9<F>(t, p.zz)));
    return a;
}

// XYZZ -> affine BE bytes (2 coords); infinity -> zeros
template <typename C>
__device__ __host__ __forceinline__ void g1_to_affine_be9(uint8_t *out, const g1jT<C> &p) {
    using F = typename C::F;
    constexpr int NB = F::W64 * 8;
    if (g1_is_inf9(p)) {
        for (int i = 0; i < 2 * F::W64; i++) ((u64 *)out)[i] = 0;
        return;
    }
    g1aT<C> a = g1_to_affine9(p);
    feT_to_be<F>(out, from_mont9<F>(a.x));
    feT_to_be<F>(out + NB, from_mont9<F>(a.y));
}

// XYZZ -> Jacobian (X_j, Y_j, Z_j) with Z_j = ZZ*ZZZ (no inversion):
//   X_j = x*Z_j^2 = X*ZZ*ZZZ^2,  Y_j = y*Z_j^3 = Y*ZZ^3*ZZZ^2
template <typename C>
__device__ __host__ __forceinline__ void g1_xyzz_to_jacobian9(feL<C::F::L> &X,
                                                     feL<C::F::L> &Y,
                                                     feL<C::F::L> &Z,
                                                     const g1jT<C> &p) {
    using F = typename C::F;
    feL<F::L> zzz2 = mont_sqr9<F>(p.zzz);
    feL<F::L> zz2 = mont_sqr9<F>(p.zz);
    X = mont_mul9<F>(mont_mul9<F>(p.x, p.zz), zzz2);
    Y = mont_mul9<F>(mont_mul9<F>(p.y, mont_mul9<F>(zz2, p.zz)), zzz2);
    Z = mont_mul9<F>(p.zz, p.zzz);
}

// Jacobian (X, Y, Z) -> XYZZ: same numerators, ZZ = Z^2, ZZZ = Z^3
template <typename C>
__device__ __host__ __forceinline__ g1jT<C> g1_jacobian_to_xyzz9(const feL<C::F::L> &X,
                                                        const feL<C::F::L> &Y,
                                                        const feL<C::F::L> &Z) {
    using F = typename C::F;
    g1jT<C> p;
    p.x = X;
    p.y = Y;
    p.zz = mont_sqr9<F>(Z);
    p.zzz = mont_mul9<F>(p.zz, Z);
    return p;
}

// wire sizes per curve (bytes); G2 specializes (4 affine / 6 jac coords)
template <typename C>
struct pt_bytes {
    static constexpr int NB = C::F::W64 * 8;  // one base-field coordinate
    static constexpr int AFF = 2 * NB;        // affine point wire
    static constexpr int JAC = 3 * NB;        // Jacobian exchange wire
};

// point negation in place (y -> -y)
template <typename C>
__device__ __host__ __forceinline__ void g1_neg_y9(g1jT<C> &p) {
    p.y = neg9<typename C::F>(p.y);
}

// ---- Jacobian wire IO (the 3-coordinate exchange payload; Z=0 = inf) ----
template <typename C>
__device__ __host__ __forceinline__ void g1_jac_be9(uint8_t *out, const g1jT<C> &p) {
    using F = typename C::F;
    constexpr int NB = F::W64 * 8;
    if (g1_is_inf9(p)) {
        for (int j = 0; j < 3 * F::W64; j++) ((u64 *)out)[j] = 0;
        return;
    }
    feL<F::L> X, Y, Z;
    g1_xyzz_to_jacobian9(X, Y, Z, p);
    feT_to_be<F>(out, from_mont9<F>(X));
    feT_to_be<F>(out + NB, from_mont9<F>(Y));
    feT_to_be<F>(out + 2 * NB, from_mont9<F>(Z));
}

// parse one Jacobian wire payload; returns false for infinity
template <typename C>
__device__ __host__ __forceinline__ bool g1_jac_from_be9(g1jT<C> &o, const uint8_t *in) {
    using F = typename C::F;
    constexpr int NB = F::W64 * 8;
    feL<F::L> X = to_mont9<F>(feT_from_be<F>(in));
    feL<F::L> Y = to_mont9<F>(feT_from_be<F>(in + NB));
    feL<F::L> Z = to_mont9<F>(feT_from_be<F>(in + 2 * NB));
    if (fe9_is_zero_modp<F>(Z)) return false;
    o = g1_jacobian_to_xyzz9<C>(X, Y, Z);
    return true;
}

}  // namespace em
