// ============================================================================
// Fq on gfx950, 9x29-bit redundant limbs (u32 each), Montgomery R' = 2^261.
//
// Why this representation (measured on MI355X, ethrex_amd/tools/int_rates):
//   v_mad_u64_u32 issues at ~full rate (~2 cyc) but 64-bit adds/moves cost
//   double, and 64-bit-limb CIOS spends over half its cycles on carry/zext
//   glue.  With 29-bit limbs every product column (up to 9 products of
//   <= 2^60 each, plus the reduction stream) fits a u64 accumulator with NO
//   intermediate carries: the multiply is 81+81 pure v_mad_u64_u32 ops.
//
// Contracts (audited in gpu_g1_9.h formulas):
//   norm2p : limbs < 2^29, value < 2p        (all stored field values)
//   mul9   : inputs limbs <= 2^30, value <= 8p; output norm2p (< 1.01p)
//   add9   : lazy limb add (no carry); out limbs <= in+1 bit, NOT norm2p
//   add9_n : lazy add + normalize + cond-sub => norm2p
//   subm9  : a - b + 2p, b a MUL OUTPUT (< 1.5p);  out norm2p
//   subn9  : a - b + 4p, b norm2p;                 out norm2p
// Borrow safety of the 2p/4p constants: oracle/gen_constants.py asserts
// c2p/c4p limb bounds against the b-value bounds above.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "bn254_constants_dev.h"

namespace em {

using u32 = uint32_t;
using u64 = uint64_t;

struct fe9 {
    u32 v[9];
};

// field traits: same 9x29 scheme for Fq (G1) and Fr (NTT)
struct Fq9T {
    static constexpr const u32 (&P)[9] = bn254::FQ9_P;
    static constexpr const u32 (&TWOP)[9] = bn254::FQ9_2P;
    static constexpr const u32 (&C2P)[9] = bn254::FQ9_C2P;
    static constexpr const u32 (&C4P)[9] = bn254::FQ9_C4P;
    static constexpr const u32 (&R2)[9] = bn254::FQ9_R2;
    static constexpr const u32 (&ONE)[9] = bn254::FQ9_ONE;
    static constexpr u32 N0INV = bn254::FQ9_N0INV;
};
struct Fr9T {
    static constexpr const u32 (&P)[9] = bn254::FR9_P;
    static constexpr const u32 (&TWOP)[9] = bn254::FR9_2P;
    static constexpr const u32 (&C2P)[9] = bn254::FR9_C2P;
    static constexpr const u32 (&C4P)[9] = bn254::FR9_C4P;
    static constexpr const u32 (&R2)[9] = bn254::FR9_R2;
    static constexpr const u32 (&ONE)[9] = bn254::FR9_ONE;
    static constexpr u32 N0INV = bn254::FR9_N0INV;
};

__device__ __host__ __forceinline__ fe9 fe9_zero() {
    return fe9{{0, 0, 0, 0, 0, 0, 0, 0, 0}};
}

__device__ __host__ __forceinline__ fe9 fe9_load(const u32 (&c)[9]) {
    fe9 r;
#pragma unroll
    for (int i = 0; i < 9; i++) r.v[i] = c[i];
    return r;
}

__device__ __host__ __forceinline__ bool fe9_eq_raw(const fe9 &a, const fe9 &b) {
    u32 d = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) d |= a.v[i] ^ b.v[i];
    return d == 0;
}

__device__ __host__ __forceinline__ bool fe9_is_zero_raw(const fe9 &a) {
    u32 d = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) d |= a.v[i];
    return d == 0;
}

// x ≡ 0 mod p for x norm2p (< 2p): x == 0 or x == p
template <typename T = Fq9T>
__device__ __forceinline__ bool fe9_is_zero_modp(const fe9 &a) {
    u32 z = 0, e = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        z |= a.v[i];
        e |= a.v[i] ^ T::P[i];
    }
    return z == 0 || e == 0;
}

// normalize limbs (ripple); input limbs < 2^32, value < 2^261
__device__ __host__ __forceinline__ fe9 fe9_norm(const fe9 &a) {
    fe9 r;
    u32 c = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        u32 t = a.v[i] + c;
        r.v[i] = t & bn254::FQ9_MASK;
        c = t >> 29;
    }
    return r;
}

// conditional subtract 2p: input norm limbs, value < 4p  =>  value < 2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 fe9_csub2p(const fe9 &a) {
    fe9 s;
    u32 bor = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        u32 t = a.v[i] - T::TWOP[i] - bor;
        bor = (t >> 31) & 1;           // limbs < 2^29 so sign bit = borrow
        s.v[i] = t & bn254::FQ9_MASK;
    }
    fe9 r;
#pragma unroll
    for (int i = 0; i < 9; i++) r.v[i] = bor ? a.v[i] : s.v[i];
    return r;
}

// conditional subtract p: input norm limbs, value < 2p  =>  canonical < p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 fe9_csubp(const fe9 &a) {
    fe9 s;
    u32 bor = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        u32 t = a.v[i] - T::P[i] - bor;
        bor = (t >> 31) & 1;
        s.v[i] = t & bn254::FQ9_MASK;
    }
    fe9 r;
#pragma unroll
    for (int i = 0; i < 9; i++) r.v[i] = bor ? a.v[i] : s.v[i];
    return r;
}

// lazy add (limbs only)
__device__ __host__ __forceinline__ fe9 add9(const fe9 &a, const fe9 &b) {
    fe9 r;
#pragma unroll
    for (int i = 0; i < 9; i++) r.v[i] = a.v[i] + b.v[i];
    return r;
}

// normalizing add: out norm2p (inputs: limb sum < 2^32, value sum < 4p)
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 add9_n(const fe9 &a, const fe9 &b) {
    return fe9_csub2p<T>(fe9_norm(add9(a, b)));
}

// a - b + 2p; b a mul output (< 1.5p, norm limbs); a norm limbs, value < 2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 subm9(const fe9 &a, const fe9 &b) {
    fe9 t;
#pragma unroll
    for (int i = 0; i < 9; i++) t.v[i] = a.v[i] + T::C2P[i] - b.v[i];
    return fe9_csub2p<T>(fe9_norm(t));    // < 4p -> < 2p
}

// a - b + 4p; b norm2p (< 2p); a norm limbs, value < 2p  => out < 6p -> 2 csubs
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 subn9(const fe9 &a, const fe9 &b) {
    fe9 t;
#pragma unroll
    for (int i = 0; i < 9; i++) t.v[i] = a.v[i] + T::C4P[i] - b.v[i];
    return fe9_csub2p<T>(fe9_csub2p<T>(fe9_norm(t)));
}

// -y mod p for y norm2p: 4p - y -> norm2p
template <typename T = Fq9T>
__device__ __forceinline__ fe9 neg9(const fe9 &y) {
    fe9 t;
#pragma unroll
    for (int i = 0; i < 9; i++) t.v[i] = T::C4P[i] - y.v[i];
    return fe9_csub2p<T>(fe9_csub2p<T>(fe9_norm(t)));
}

// ---- Montgomery multiplication: column SOS, radix 2^29 ----
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 mont_mul9(const fe9 &A, const fe9 &B) {
    u64 t[17];
#pragma unroll
    for (int k = 0; k < 17; k++) t[k] = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
#pragma unroll
        for (int j = 0; j < 9; j++) t[i + j] += (u64)A.v[i] * B.v[j];
    }
#pragma unroll
    for (int k = 0; k < 9; k++) {
        u32 m = ((u32)t[k] * T::N0INV) & bn254::FQ9_MASK;
#pragma unroll
        for (int j = 0; j < 9; j++) t[k + j] += (u64)m * T::P[j];
        t[k + 1] += t[k] >> 29;        // t[k] ≡ 0 mod 2^29 now
    }
    fe9 r;
    u64 c = 0;
#pragma unroll
    for (int k = 9; k < 17; k++) {
        c += t[k];
        r.v[k - 9] = (u32)c & bn254::FQ9_MASK;
        c >>= 29;
    }
    r.v[8] = (u32)c;                   // result < 1.01p => fits 29 bits
    return r;
}

template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 mont_sqr9(const fe9 &a) {
    return mont_mul9<T>(a, a);
}

// ---- conversions fe4 (4x64 canonical) <-> fe9 ----

// canonical (or any < 2^256) u64[4] -> 29-bit limbs (raw, norm limbs)
__device__ __host__ __forceinline__ fe9 fe9_from_u64x4(const u64 w[4]) {
    fe9 r;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        int bit = 29 * i;
        int word = bit >> 6, off = bit & 63;
        u64 lo = w[word] >> off;
        if (off > 35 && word < 3) lo |= w[word + 1] << (64 - off);
        r.v[i] = (u32)lo & bn254::FQ9_MASK;
    }
    return r;
}

// canonical fe9 (< p, norm limbs) -> u64[4]
__device__ __host__ __forceinline__ void fe9_to_u64x4(u64 w[4], const fe9 &a) {
#pragma unroll
    for (int i = 0; i < 4; i++) w[i] = 0;
#pragma unroll
    for (int i = 0; i < 9; i++) {
        int bit = 29 * i;
        int word = bit >> 6, off = bit & 63;
        w[word] |= (u64)a.v[i] << off;
        if (off > 35 && word < 3) w[word + 1] |= (u64)a.v[i] >> (64 - off);
    }
}

// to Montgomery(2^261): x any value < 2^256 (raw 29-limbs) -> norm2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 to_mont9(const fe9 &x) {
    return mont_mul9<T>(x, fe9_load(T::R2));
}

// from Montgomery: norm2p -> canonical (< p, norm limbs)
template <typename T = Fq9T>
__device__ __host__ __forceinline__ fe9 from_mont9(const fe9 &x) {
    fe9 one{{1, 0, 0, 0, 0, 0, 0, 0, 0}};
    return fe9_csubp<T>(mont_mul9<T>(x, one));
}

// x^e (Montgomery in/out), e canonical 4x64
template <typename T = Fq9T>
__device__ __forceinline__ fe9 mont_pow9(const fe9 &x, const u64 e[4]) {
    fe9 acc = fe9_load(T::ONE);
    for (int i = 255; i >= 0; i--) {
        acc = mont_sqr9<T>(acc);
        if ((e[i >> 6] >> (i & 63)) & 1) acc = mont_mul9<T>(acc, x);
    }
    return acc;
}

// 1/x via Fermat (x norm2p, != 0 mod p)
__device__ __forceinline__ fe9 mont_inv9(const fe9 &x) {
    u64 e[4] = {bn254::Fq::MOD[0] - 2, bn254::Fq::MOD[1], bn254::Fq::MOD[2],
                bn254::Fq::MOD[3]};
    return mont_pow9(x, e);
}

// raw lexicographic compare of norm-limb values: a >= b ?
__device__ __forceinline__ bool fe9_geq_raw(const fe9 &a, const u32 (&b)[9]) {
#pragma unroll
    for (int i = 8; i >= 0; i--) {
        if (a.v[i] != b[i]) return a.v[i] > b[i];
    }
    return true;
}

// equality mod p of two norm2p values (canonicalize then compare)
template <typename T = Fq9T>
__device__ __forceinline__ bool fe9_eq_modp(const fe9 &a, const fe9 &b) {
    return fe9_eq_raw(fe9_csubp<T>(a), fe9_csubp<T>(b));
}

}  // namespace em
