// ============================================================================
// Prime fields on gfx950: L x 29-bit redundant limbs (u32 each), Montgomery
// R' = 2^(29L).  Instances: BN254 Fq/Fr (L=9, R'=2^261) and BLS12-381 Fp
// (L=14, R'=2^406) + its Fr on the 4x64 digit path.
//
// Why this representation (measured on MI355X, ethrex_amd/tools/int_rates):
//   v_mad_u64_u32 issues at ~full rate (~2 cyc) but 64-bit adds/moves cost
//   double, and 64-bit-limb CIOS spends over half its cycles on carry/zext
//   glue.  With 29-bit limbs every product column fits a u64 accumulator
//   with NO intermediate carries: the multiply is L^2+L^2 pure
//   v_mad_u64_u32 ops (81+81 at L=9 => 256 instructions total).
//
// Column-overflow budget: column <= (2L)*(limb_max)^2 + carries < 2^64.
//   L=9:  limbs <= 2^30 allowed (one lazy-add level may feed a mul).
//   L=14: limbs <= 2^29 REQUIRED on mul inputs (28*2^58 < 2^63; lazy adds
//         must be normalized first — the curve layer uses add9_n there).
//
// Contracts (audited at every call site in gpu_g1_9.h):
//   norm2p : limbs < 2^29, value < 2p        (all stored field values)
//   mul    : inputs per the L-rule above, value <= 8p; output norm2p (<1.01p)
//   add9   : lazy limb add (no carry); NOT norm2p (L=9 mul inputs only)
//   add9_n : lazy add + normalize + cond-sub => norm2p
//   subm9  : a - b + 2p, b a MUL OUTPUT (< 1.5p);  out norm2p
//   subn9  : a - b + 4p, b norm2p;                 out norm2p
// Borrow safety of the 2p/4p constants is asserted against the b-value
// bounds in oracle/gen_constants.py for both fields.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "bn254_constants_dev.h"

namespace em {

using u32 = uint32_t;
using u64 = uint64_t;

template <int LN>
struct feL {
    u32 v[LN];
};

using fe9 = feL<9>;    // BN254 Fq and Fr share the layout
using fe14 = feL<14>;  // BLS12-381 Fp

// field traits
struct Fq9T {
    static constexpr int L = 9;
    static constexpr int W64 = 4;  // 64-bit words of canonical IO
    static constexpr const u32 (&P)[9] = bn254::FQ9_P;
    static constexpr const u32 (&TWOP)[9] = bn254::FQ9_2P;
    static constexpr const u32 (&C2P)[9] = bn254::FQ9_C2P;
    static constexpr const u32 (&C4P)[9] = bn254::FQ9_C4P;
    static constexpr const u32 (&R2)[9] = bn254::FQ9_R2;
    static constexpr const u32 (&ONE)[9] = bn254::FQ9_ONE;
    static constexpr const u64 (&MOD64)[4] = bn254::Fq::MOD;
    static constexpr u32 N0INV = bn254::FQ9_N0INV;
};
struct Fr9T {
    static constexpr int L = 9;
    static constexpr int W64 = 4;
    static constexpr const u32 (&P)[9] = bn254::FR9_P;
    static constexpr const u32 (&TWOP)[9] = bn254::FR9_2P;
    static constexpr const u32 (&C2P)[9] = bn254::FR9_C2P;
    static constexpr const u32 (&C4P)[9] = bn254::FR9_C4P;
    static constexpr const u32 (&R2)[9] = bn254::FR9_R2;
    static constexpr const u32 (&ONE)[9] = bn254::FR9_ONE;
    static constexpr const u64 (&MOD64)[4] = bn254::Fr::MOD;
    static constexpr u32 N0INV = bn254::FR9_N0INV;
};
struct FpB14T {
    static constexpr int L = 14;
    static constexpr int W64 = 6;
    static constexpr const u32 (&P)[14] = bn254::FPB_P;
    static constexpr const u32 (&TWOP)[14] = bn254::FPB_2P;
    static constexpr const u32 (&C2P)[14] = bn254::FPB_C2P;
    static constexpr const u32 (&C4P)[14] = bn254::FPB_C4P;
    static constexpr const u32 (&R2)[14] = bn254::FPB_R2;
    static constexpr const u32 (&ONE)[14] = bn254::FPB_ONE;
    static constexpr const u64 (&MOD64)[6] = bn254::FPB_MOD64;
    static constexpr u32 N0INV = bn254::FPB_N0INV;
};

template <int LN>
__device__ __host__ __forceinline__ feL<LN> fe9z() {
    feL<LN> r;
#pragma unroll
    for (int i = 0; i < LN; i++) r.v[i] = 0;
    return r;
}

__device__ __host__ __forceinline__ fe9 fe9_zero() { return fe9z<9>(); }

template <int LN>
__device__ __host__ __forceinline__ feL<LN> fe9_load(const u32 (&c)[LN]) {
    feL<LN> r;
#pragma unroll
    for (int i = 0; i < LN; i++) r.v[i] = c[i];
    return r;
}

template <int LN>
__device__ __host__ __forceinline__ bool fe9_eq_raw(const feL<LN> &a,
                                                    const feL<LN> &b) {
    u32 d = 0;
#pragma unroll
    for (int i = 0; i < LN; i++) d |= a.v[i] ^ b.v[i];
    return d == 0;
}

template <int LN>
__device__ __host__ __forceinline__ bool fe9_is_zero_raw(const feL<LN> &a) {
    u32 d = 0;
#pragma unroll
    for (int i = 0; i < LN; i++) d |= a.v[i];
    return d == 0;
}

// x ≡ 0 mod p for x norm2p (< 2p): x == 0 or x == p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ bool fe9_is_zero_modp(const feL<T::L> &a) {
    u32 z = 0, e = 0;
#pragma unroll
    for (int i = 0; i < T::L; i++) {
        z |= a.v[i];
        e |= a.v[i] ^ T::P[i];
    }
    return z == 0 || e == 0;
}

// normalize limbs (ripple); input limbs < 2^32, value < 2^(29L)
template <int LN>
__device__ __host__ __forceinline__ feL<LN> fe9_norm(const feL<LN> &a) {
    feL<LN> r;
    u32 c = 0;
#pragma unroll
    for (int i = 0; i < LN; i++) {
        u32 t = a.v[i] + c;
        r.v[i] = t & bn254::FQ9_MASK;
        c = t >> 29;
    }
    return r;
}

// conditional subtract 2p: input norm limbs, value < 4p  =>  value < 2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> fe9_csub2p(const feL<T::L> &a) {
    feL<T::L> s;
    u32 bor = 0;
#pragma unroll
    for (int i = 0; i < T::L; i++) {
        u32 t = a.v[i] - T::TWOP[i] - bor;
        bor = (t >> 31) & 1;           // limbs < 2^29 so sign bit = borrow
        s.v[i] = t & bn254::FQ9_MASK;
    }
    feL<T::L> r;
#pragma unroll
    for (int i = 0; i < T::L; i++) r.v[i] = bor ? a.v[i] : s.v[i];
    return r;
}

// conditional subtract p: input norm limbs, value < 2p  =>  canonical < p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> fe9_csubp(const feL<T::L> &a) {
    feL<T::L> s;
    u32 bor = 0;
#pragma unroll
    for (int i = 0; i < T::L; i++) {
        u32 t = a.v[i] - T::P[i] - bor;
        bor = (t >> 31) & 1;
        s.v[i] = t & bn254::FQ9_MASK;
    }
    feL<T::L> r;
#pragma unroll
    for (int i = 0; i < T::L; i++) r.v[i] = bor ? a.v[i] : s.v[i];
    return r;
}

// lazy add (limbs only; L=9 may feed a mul directly, L=14 may NOT)
template <int LN>
__device__ __host__ __forceinline__ feL<LN> add9(const feL<LN> &a,
                                                 const feL<LN> &b) {
    feL<LN> r;
#pragma unroll
    for (int i = 0; i < LN; i++) r.v[i] = a.v[i] + b.v[i];
    return r;
}

// normalizing add: out norm2p (inputs: limb sum < 2^32, value sum < 4p)
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> add9_n(const feL<T::L> &a,
                                                     const feL<T::L> &b) {
    return fe9_csub2p<T>(fe9_norm<T::L>(add9<T::L>(a, b)));
}

// a - b + 2p; b a mul output (< 1.5p, norm limbs); a norm limbs, value < 2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> subm9(const feL<T::L> &a,
                                                    const feL<T::L> &b) {
    feL<T::L> t;
#pragma unroll
    for (int i = 0; i < T::L; i++) t.v[i] = a.v[i] + T::C2P[i] - b.v[i];
    return fe9_csub2p<T>(fe9_norm<T::L>(t));    // < 4p -> < 2p
}

// a - b + 4p; b norm2p (< 2p); a norm limbs, value < 2p => out < 6p -> 2 csubs
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> subn9(const feL<T::L> &a,
                                                    const feL<T::L> &b) {
    feL<T::L> t;
#pragma unroll
    for (int i = 0; i < T::L; i++) t.v[i] = a.v[i] + T::C4P[i] - b.v[i];
    return fe9_csub2p<T>(fe9_csub2p<T>(fe9_norm<T::L>(t)));
}

// -y mod p for y norm2p: 4p - y -> norm2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> neg9(const feL<T::L> &y) {
    feL<T::L> t;
#pragma unroll
    for (int i = 0; i < T::L; i++) t.v[i] = T::C4P[i] - y.v[i];
    return fe9_csub2p<T>(fe9_csub2p<T>(fe9_norm<T::L>(t)));
}

// ---- Montgomery multiplication: column SOS, radix 2^29 ----
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> mont_mul9(const feL<T::L> &A,
                                                        const feL<T::L> &B) {
    constexpr int LN = T::L;
    u64 t[2 * LN - 1];
#pragma unroll
    for (int k = 0; k < 2 * LN - 1; k++) t[k] = 0;
#pragma unroll
    for (int i = 0; i < LN; i++) {
#pragma unroll
        for (int j = 0; j < LN; j++) t[i + j] += (u64)A.v[i] * B.v[j];
    }
#pragma unroll
    for (int k = 0; k < LN; k++) {
        u32 m = ((u32)t[k] * T::N0INV) & bn254::FQ9_MASK;
#pragma unroll
        for (int j = 0; j < LN; j++) t[k + j] += (u64)m * T::P[j];
        t[k + 1] += t[k] >> 29;        // t[k] ≡ 0 mod 2^29 now
    }
    feL<LN> r;
    u64 c = 0;
#pragma unroll
    for (int k = LN; k < 2 * LN - 1; k++) {
        c += t[k];
        r.v[k - LN] = (u32)c & bn254::FQ9_MASK;
        c >>= 29;
    }
    r.v[LN - 1] = (u32)c;              // result < 1.01p => fits 29 bits
    return r;
}

// Montgomery squaring: the 81 product mads shrink to 45 (36 cross terms
// with pre-doubled multiplicands + 9 diagonals) — ~11% fewer instructions
// per square; 2 of the ~10 muls in a mixed point add are squares.  Column
// values are IDENTICAL to mont_mul9(a,a) (partial sums are positive and
// bounded by the final column value), so the overflow budget is unchanged;
// d[i] = 2*a[i] <= 2^31 fits u32 for both the L=9 (inputs <= 2^30) and
// L=14 (inputs < 2^29) rules.
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> mont_sqr9(const feL<T::L> &a) {
    constexpr int LN = T::L;
    u64 t[2 * LN - 1];
#pragma unroll
    for (int k = 0; k < 2 * LN - 1; k++) t[k] = 0;
    u32 d[LN];
#pragma unroll
    for (int i = 0; i < LN; i++) d[i] = a.v[i] << 1;
#pragma unroll
    for (int i = 0; i < LN; i++) {
        t[2 * i] += (u64)a.v[i] * a.v[i];
#pragma unroll
        for (int j = i + 1; j < LN; j++) t[i + j] += (u64)d[i] * a.v[j];
    }
#pragma unroll
    for (int k = 0; k < LN; k++) {
        u32 m = ((u32)t[k] * T::N0INV) & bn254::FQ9_MASK;
#pragma unroll
        for (int j = 0; j < LN; j++) t[k + j] += (u64)m * T::P[j];
        t[k + 1] += t[k] >> 29;
    }
    feL<LN> r;
    u64 c = 0;
#pragma unroll
    for (int k = LN; k < 2 * LN - 1; k++) {
        c += t[k];
        r.v[k - LN] = (u32)c & bn254::FQ9_MASK;
        c >>= 29;
    }
    r.v[LN - 1] = (u32)c;
    return r;
}

// ---- conversions canonical u64 words <-> 29-bit limbs ----

template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> fe9_from_u64(const u64 *w) {
    feL<T::L> r;
#pragma unroll
    for (int i = 0; i < T::L; i++) {
        int bit = 29 * i;
        int word = bit >> 6, off = bit & 63;
        u64 lo = w[word] >> off;
        if (off > 35 && word < T::W64 - 1) lo |= w[word + 1] << (64 - off);
        r.v[i] = (u32)lo & bn254::FQ9_MASK;
    }
    return r;
}

template <typename T = Fq9T>
__device__ __host__ __forceinline__ void fe9_to_u64(u64 *w, const feL<T::L> &a) {
#pragma unroll
    for (int i = 0; i < T::W64; i++) w[i] = 0;
#pragma unroll
    for (int i = 0; i < T::L; i++) {
        int bit = 29 * i;
        int word = bit >> 6, off = bit & 63;
        w[word] |= (u64)a.v[i] << off;
        if (off > 35 && word < T::W64 - 1) w[word + 1] |= (u64)a.v[i] >> (64 - off);
    }
}

// legacy names used by the BN254 G1/NTT code (4x64 IO)
__device__ __host__ __forceinline__ fe9 fe9_from_u64x4(const u64 w[4]) {
    return fe9_from_u64<Fq9T>(w);
}
__device__ __host__ __forceinline__ void fe9_to_u64x4(u64 w[4], const fe9 &a) {
    fe9_to_u64<Fq9T>(w, a);
}

// to Montgomery: x any canonical-width value (raw 29-limbs) -> norm2p
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> to_mont9(const feL<T::L> &x) {
    return mont_mul9<T>(x, fe9_load<T::L>(T::R2));
}

// from Montgomery: norm2p -> canonical (< p, norm limbs)
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> from_mont9(const feL<T::L> &x) {
    feL<T::L> one = fe9z<T::L>();
    one.v[0] = 1;
    return fe9_csubp<T>(mont_mul9<T>(x, one));
}

// x^e (Montgomery in/out), e canonical u64[T::W64]
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> mont_pow9(const feL<T::L> &x, const u64 *e) {
    feL<T::L> acc = fe9_load<T::L>(T::ONE);
    for (int i = 64 * T::W64 - 1; i >= 0; i--) {
        acc = mont_sqr9<T>(acc);
        if ((e[i >> 6] >> (i & 63)) & 1) acc = mont_mul9<T>(acc, x);
    }
    return acc;
}

// 1/x via Fermat (x norm2p, != 0 mod p); exponent p-2 (p odd => no borrow)
template <typename T = Fq9T>
__device__ __host__ __forceinline__ feL<T::L> mont_inv9(const feL<T::L> &x) {
    u64 e[T::W64];
#pragma unroll
    for (int i = 0; i < T::W64; i++) e[i] = T::MOD64[i];
    e[0] -= 2;
    return mont_pow9<T>(x, e);
}

// raw lexicographic compare of norm-limb values: a >= b ?
template <int LN>
__device__ __forceinline__ bool fe9_geq_raw(const feL<LN> &a, const u32 (&b)[LN]) {
#pragma unroll
    for (int i = LN - 1; i >= 0; i--) {
        if (a.v[i] != b[i]) return a.v[i] > b[i];
    }
    return true;
}

// equality mod p of two norm2p values (canonicalize then compare)
template <typename T = Fq9T>
__device__ __forceinline__ bool fe9_eq_modp(const feL<T::L> &a,
                                            const feL<T::L> &b) {
    return fe9_eq_raw<T::L>(fe9_csubp<T>(a), fe9_csubp<T>(b));
}

}  // namespace em
