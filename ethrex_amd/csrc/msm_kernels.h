// ============================================================================
// Pippenger bucket MSM — gfx950 kernels, curve- and geometry-templated
// (BN254 G1, BLS12-381 G1 and G2-over-Fp2 on the 29-bit-limb field core).
//
// Pipeline per run (window bits CB from msm_cfg; SIGNED c=17 for large
// BN254 MSMs, unsigned c=16 for large BLS, c=8 for n <= 2^16, FB_C=13
// fixed-base for blob-KZG):
//   1. parse scalars (BN254: ark from_be_bytes_mod_order reduction;
//      BLS: raw 256-bit per blst — both as packed u64[4] for digits)
//   2. digit extraction -> (key, value = point index [+ sign/skip bits
//      for signed configs]); keys are u16 in-window ids for the
//      per-window-sort path (n >= 2^23), u32 (window<<IB | id) otherwise
//   3. rocPRIM radix sorts: one per window over just the id bits (large
//      n), or one global sort (small n — ~60 fewer host launches/step)
//   4. per-bucket offsets by binary search + LENGTH-SORTED bucket schedule
//      (waves then process similar-length runs: no Poisson divergence)
//   5. bucket accumulation: one thread per scheduled bucket walks its run
//      with XYZZ+affine mixed adds (negating y for negative digits) —
//      the VALU-issue-bound hot kernel (saturated at 3 waves/SIMD:
//      profiles/r02_summary.md)
//   6. segment running sums -> weighted reduce -> per-window LDS trees ->
//      k_emit_windows.  Windows leave UNSCALED; the host folds the
//      2^(CB*w) factors (Horner) at delivery, overlapped with the next
//      pipelined step's GPU work.
//
// Work shape: ~NWIN*(n + 2^(IB+1)) mixed adds; HBM traffic is only the
// gathered points + sorted pairs => VALU-bound, so there
// is deliberately no MFMA anywhere here.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "gpu_field.h"   // fe4: Fr scalar handling
#include "gpu_g1_9.h"    // fe9: Fq / G1 compute core
#include "gpu_g2.h"      // fp2: BLS12-381 G2 specializations

namespace em {

// Window geometry is a compile-time config: CB = window bits.  Large MSMs
// use c=16 (bucket work dominates); small (blob-KZG-sized) MSMs use c=8 so
// the fixed bucket-reduction tail shrinks 256x (measured 7.3 ms -> ~2 ms
// per 4096-point commitment).  Scalar width: BN254 scalars are reduced mod
// r (254 bits); BLS12-381 scalars are raw 256-bit integers.
// SGN = signed (balanced) digits: window values are recoded to
// d in [-2^(CB-1), +2^(CB-1)] with a carry into the next window, so each
// window needs only 2^(CB-1) buckets (id = |d|-1; the sign rides in the
// sorted value's bit 31 and the bucket walk negates y on the fly — point
// negation is free on a short-Weierstrass curve).  That allows CB=17 for
// BN254: 15 windows instead of 16 (6% fewer point adds, the VALU-issue-
// bound budget) at the same bucket count as unsigned c=16.  Requires the
// top window to absorb the final carry: SBITS % CB != 0, and scalars
// reduced (BN254 only; BLS 256-bit raw scalars stay unsigned).
template <int CB, int SBITS, bool SGN = false>
struct msm_cfg {
    static constexpr bool SIGNED = SGN;
    static_assert(!SGN || SBITS % CB != 0, "top window must absorb carry");
    static constexpr int C = CB;               // window bits (digit extract)
    static constexpr int IB = SGN ? CB - 1 : CB;  // bucket-index bits
    static constexpr int NWIN = (SBITS + CB - 1) / CB;
    static constexpr uint32_t DMASK = (1u << IB) - 1;
    static constexpr uint32_t NBUCKETS = (uint32_t)NWIN << IB;
    static constexpr int DBITS = IB;           // per-window sort key bits
    static constexpr int SORT_BITS = IB + 6;   // + window bits (global sort)
    // SEG tuned per window size (GPU-measured): 16 for the 64K-bucket c=16
    // windows (SEG=8 doubles weighted-reduce work there, reduce 1.8->2.5 ms);
    // 2 for the smaller FB/c=8 windows — their reduce is LATENCY-bound on
    // few waves (G2: 1024 threads total), so short segment scans + 4x the
    // threads beat longer per-thread runs (r01 had 8; r02 host Horner
    // removed the window-scale chains that dominated before)
    static constexpr int SEG = (1 << IB) >= 65536 ? 16 : 2;
    static constexpr int NSEG = (1 << IB) / SEG;        // segments per window
    static constexpr int RED_BLOCK = 256;
    // when a 256-thread block spans multiple windows the LDS tree is skipped
    static constexpr bool TREE = NSEG >= RED_BLOCK;
    static constexpr int NPART = TREE ? NWIN * (NSEG / RED_BLOCK) : NWIN * NSEG;
};

// the shipped geometries (both curves)
using CfgL254 = msm_cfg<16, 254>;      // BN254 large unsigned (tree mode)
using CfgSg254 = msm_cfg<17, 254, true>;  // BN254 large DEFAULT: signed c=17
using CfgS254 = msm_cfg<8, 254>;       // BN254 small (n <= 2^16)
using CfgL256 = msm_cfg<16, 256>;      // BLS large
using CfgS256 = msm_cfg<8, 256>;       // BLS small

// sorted-value bit layout for SIGNED configs (vals[] entries)
constexpr uint32_t SGN_NEG = 0x80000000u;   // digit is negative: add -P
constexpr uint32_t SGN_SKIP = 0x40000000u;  // zero digit / infinity: no add
constexpr uint32_t SGN_IDX = 0x3fffffffu;   // point index (n < 2^30)

// digit w = bits [CB*w, CB*w+CB) of the scalar (spans u64 limbs)
template <int CB>
__device__ __forceinline__ uint32_t msm_digit(const fe4 &k, int w) {
    int bit = CB * w;
    int limb = bit >> 6, off = bit & 63;
    uint64_t d = k.v[limb] >> off;
    if (off > 64 - CB && limb < 3) d |= k.v[limb + 1] << (64 - off);
    return (uint32_t)d & ((1u << CB) - 1);
}

// ---- input parsing ----

// 64-byte BE affine -> Montgomery(2^261) g1a9 + infinity flag; off-curve ->
// err.  Coordinates reduced mod p, (0,0) = identity (provider.rs:252-268).
static __global__ void k_parse_points(const uint8_t *__restrict__ in,
                               g1a9 *__restrict__ pts, uint8_t *__restrict__ inf,
                               size_t n, uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe9 x = to_mont9(fe9_from_be(in + 64 * i));
    fe9 y = to_mont9(fe9_from_be(in + 64 * i + 32));
    if (fe9_is_zero_modp(x) && fe9_is_zero_modp(y)) {
        inf[i] = 1;
        pts[i].x = fe9_zero();
        pts[i].y = fe9_zero();
        return;
    }
    g1a9 p{x, y};
    inf[i] = 0;
    if (!g1a9_on_curve(p)) atomicOr(err, 1u);
    pts[i] = p;
}

// P_i = (start+i+1)*G directly in HBM; per-thread affine conversion.
static __global__ void k_gen_points(g1a9 *__restrict__ pts, uint8_t *__restrict__ inf,
                             size_t n, uint64_t start) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    u64 k[4] = {start + i + 1, 0, 0, 0};
    g1a9 g = g1_generator9();
    g1j9 acc = g1_inf9();
    for (int b = 63; b >= 0; b--) {
        acc = g1_dbl9(acc);
        if ((k[0] >> b) & 1) acc = g1_add_affine9(acc, g);
    }
    // to affine (k >= 1 and k < r => never infinity)
    pts[i] = g1_to_affine9(acc);
    inf[i] = 0;
}

// download points as BE affine bytes (for parity tests)
static __global__ void k_points_to_be(const g1a9 *__restrict__ pts,
                               const uint8_t *__restrict__ inf,
                               uint8_t *__restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (inf[i]) {
        for (int j = 0; j < 8; j++) ((u64 *)(out + 64 * i))[j] = 0;
        return;
    }
    fe9_to_be(out + 64 * i, from_mont9(pts[i].x));
    fe9_to_be(out + 64 * i + 32, from_mont9(pts[i].y));
}

// 32-byte BE scalars -> canonical fe4 reduced mod r (from_be_bytes_mod_order)
static __global__ void k_parse_scalars(const uint8_t *__restrict__ in,
                                fe4 *__restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    out[i] = from_mont<Fr>(to_mont<Fr>(fe_from_be(in + 32 * i)));
}

// ---- digit extraction ----
// UNSIGNED: key = (w << C) | digit, identity points park in digit-0 buckets
// (skipped by the walk).  SIGNED: balanced recode with carry — window value
// t = raw + carry_in; t <= 2^(C-1) keeps d = +t, else d = t - 2^C (carry 1);
// bucket id = |d| - 1 (zero digits / infinities park at id 0 with SGN_SKIP).
template <typename CFG, typename KT = uint32_t>
static __global__ void k_digits(const fe4 *__restrict__ scalars,
                         const uint8_t *__restrict__ inf,
                         KT *__restrict__ keys, uint32_t *__restrict__ vals,
                         size_t n) {
    // u16 keys: in-window id only (window implicit by segment; 25% less
    // sort traffic).  u32 keys: (w << IB/C) | id, globally sortable.
    constexpr bool K16 = sizeof(KT) == 2;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe4 k = scalars[i];
    bool skip = inf[i];
    if constexpr (CFG::SIGNED) {
        uint32_t carry = 0;
#pragma unroll
        for (int w = 0; w < CFG::NWIN; w++) {
            uint32_t t = msm_digit<CFG::C>(k, w) + carry;
            uint32_t sign = 0, mag = t;
            if (t > (1u << (CFG::C - 1))) {
                mag = (1u << CFG::C) - t;
                sign = 1;
                carry = 1;
            } else {
                carry = 0;
            }
            bool sk = skip || mag == 0;
            uint32_t id = sk ? 0u : mag - 1;
            keys[(size_t)w * n + i] =
                (KT)(K16 ? id : (((uint32_t)w << CFG::IB) | id));
            vals[(size_t)w * n + i] =
                (uint32_t)i | (sign << 31) | (sk ? SGN_SKIP : 0u);
        }
        // scalars are reduced mod r < 2^254 and 254 % 17 = 16, so the top
        // window value (+ carry) is <= 2^16 = 2^(C-1): never a carry out
    } else {
#pragma unroll
        for (int w = 0; w < CFG::NWIN; w++) {
            uint32_t d = msm_digit<CFG::C>(k, w);
            if (skip) d = 0;  // identity points: park in bucket 0
            keys[(size_t)w * n + i] =
                (KT)(K16 ? d : (((uint32_t)w << CFG::C) | d));
            vals[(size_t)w * n + i] = (uint32_t)i;
        }
    }
}

// segment-local offsets for u16 per-window keys: bucket b = (w, id); its
// run is the lower_bound of id within the window's segment [w*n, (w+1)*n)
template <typename CFG>
static __global__ void k_offsets_seg(const uint16_t *__restrict__ sorted_keys,
                                     size_t n /* per window */,
                                     uint32_t *__restrict__ offsets) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b > CFG::NBUCKETS) return;
    if (b == CFG::NBUCKETS) {
        offsets[b] = (uint32_t)(n * CFG::NWIN);
        return;
    }
    uint32_t w = b >> CFG::IB;
    uint16_t id = (uint16_t)(b & CFG::DMASK);
    const uint16_t *seg = sorted_keys + (size_t)w * n;
    size_t lo = 0, hi = n;
    while (lo < hi) {
        size_t mid = (lo + hi) >> 1;
        if (seg[mid] < id)
            lo = mid + 1;
        else
            hi = mid;
    }
    offsets[b] = (uint32_t)((size_t)w * n + lo);
}

// ---- bucket segment offsets: lower_bound of each bucket id ----
template <typename CFG>
static __global__ void k_offsets(const uint32_t *__restrict__ sorted_keys, size_t total,
                          uint32_t *__restrict__ offsets) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b > CFG::NBUCKETS) return;
    if (b == CFG::NBUCKETS) {
        offsets[b] = (uint32_t)total;
        return;
    }
    size_t lo = 0, hi = total;
    while (lo < hi) {
        size_t mid = (lo + hi) >> 1;
        if (sorted_keys[mid] < b)
            lo = mid + 1;
        else
            hi = mid;
    }
    offsets[b] = (uint32_t)lo;
}

// ---- bucket-length schedule: sort bucket ids by run length so each wave
// processes similar-length runs (a wave pays max-of-64 Poisson run lengths
// otherwise: measured ~1.2-1.5x divergence loss) ----
template <typename CFG>
static __global__ void k_bucket_lengths(const uint32_t *__restrict__ offsets,
                                 uint32_t *__restrict__ len,
                                 uint32_t *__restrict__ ids) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= CFG::NBUCKETS) return;
    // unsigned: digit-0 buckets are dead (identity parking); signed: all live
    len[b] = (!CFG::SIGNED && (b & CFG::DMASK) == 0)
                 ? 0u
                 : offsets[b + 1] - offsets[b];
    ids[b] = b;
}

// ---- bucket accumulation (the hot kernel) ----
// one thread per SCHEDULED bucket id; digit-0 buckets skipped.
// GATHER=true reads pts[vals[t]] (sorted-pair gather); GATHER=false reads
// the pairing-tree level buffer directly (vals may be null then).
template <typename C, typename CFG, bool GATHER = true>
__global__ void __launch_bounds__(256)
k_bucket_acc(const g1aT<C> *__restrict__ pts, const uint32_t *__restrict__ vals,
             const uint32_t *__restrict__ offsets,
             const uint32_t *__restrict__ sched, g1jT<C> *__restrict__ buckets,
             uint32_t tid_base = 0) {
    uint32_t tid = blockIdx.x * blockDim.x + threadIdx.x + tid_base;
    if (tid >= CFG::NBUCKETS) return;
    uint32_t b = sched[tid];
    if constexpr (!CFG::SIGNED) {
        if ((b & CFG::DMASK) == 0) return;  // digit 0
    }
    uint32_t lo = offsets[b], hi = offsets[b + 1];
    g1jT<C> acc = g1_inf9<C>();
    if (lo >= hi) {
        buckets[b] = acc;
        return;
    }
    constexpr uint32_t IMASK = CFG::SIGNED ? SGN_IDX : 0xffffffffu;
    // software pipeline: issue the NEXT point's gather before the long mixed
    // add so the dependent idx->point load chain overlaps the VALU work.
    uint32_t v = GATHER ? vals[lo] : 0;
    g1aT<C> p = GATHER ? pts[v & IMASK] : pts[lo];
    for (uint32_t t = lo; t < hi; t++) {
        g1aT<C> cur = p;
        uint32_t vc = v;
        uint32_t nxt = t + 1 < hi ? t + 1 : t;
        v = GATHER ? vals[nxt] : 0;
        p = GATHER ? pts[v & IMASK] : pts[nxt];
        if constexpr (CFG::SIGNED) {
            if (vc & SGN_SKIP) continue;  // parked zero digit / infinity
            if (vc & SGN_NEG) cur.y = neg9<typename C::F>(cur.y);  // add -P
        }
        if constexpr (!GATHER) {
            // tree outputs may be the identity (P + (-P)): encoded (0,0)
            if (fe9_is_zero_raw<C::F::L>(cur.x) &&
                fe9_is_zero_raw<C::F::L>(cur.y))
                continue;
        }
        acc = g1_add_affine9(acc, cur);
    }
    buckets[b] = acc;
}

// ---- batch-affine pairing tree (large BN254 MSMs) ----
// The XYZZ mixed add above costs ~13 Montgomery muls per point add.  For
// large MSMs the bucket runs are long (avg n/2^16 = 256 at 2^24), so the
// bucket sums can instead be built as a level-synchronized pairing tree of
// AFFINE adds: lambda = (y2-y1)/(x2-x1), x3 = lambda^2-x1-x2,
// y3 = lambda(x1-x3)-y1 — ~6 muls per add once the inversion is batched.
// Each thread processes PAIR_K consecutive pair slots per level: a forward
// pass chains denominator prefix-products (Montgomery's trick, products
// stored to an aux buffer), ONE Fermat inversion per thread (amortized
// ~1.2 muls/add at PAIR_K=256, and fully parallel across threads), then a
// backward pass peels per-pair inverses and writes the sums.  Identity /
// doubling / P+(-P) / odd-singleton cases ride the same batch with an
// identity denominator (doubling contributes its real 2y denominator, so
// it needs no separate inversion).  After the tree levels shrink runs to
// <= ~4, the XYZZ kernel (GATHER=false) finishes against the level buffer.
#ifndef EM_PAIR_K
#define EM_PAIR_K 256
#endif
constexpr int PAIR_K = EM_PAIR_K;  // pair slots per thread per level

// per-bucket pair counts for the next level: ceil(len/2); digit-0 buckets
// contribute nothing (skip_d0 set at level 0; empty thereafter).
static __global__ void k_pair_counts(const uint32_t *__restrict__ off_in,
                              uint32_t *__restrict__ cnt, uint32_t nbuckets,
                              uint32_t dmask, int skip_d0) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b > nbuckets) return;
    if (b == nbuckets) {
        cnt[b] = 0;
        return;
    }
    if (skip_d0 && (b & dmask) == 0) {
        cnt[b] = 0;
        return;
    }
    cnt[b] = (off_in[b + 1] - off_in[b] + 1) >> 1;
}

// classification of one pair slot; d/n are only meaningful when kind==0
template <typename C>
struct pair_case_t {
    feL<C::F::L> d, n;   // lambda = n / d
    g1aT<C> copy;        // kind==1 result
    int kind;            // 0 = lambda path, 1 = copy/identity result
};

template <typename C>
__device__ __forceinline__ bool g1a_is_inf(const g1aT<C> &p) {
    using T = typename C::F;
    return fe9_is_zero_modp<T>(p.x) && fe9_is_zero_modp<T>(p.y);
}

// classify pair (A,B); sgl means B absent (odd tail)
template <typename C>
__device__ __forceinline__ pair_case_t<C> pair_classify(const g1aT<C> &A,
                                                        const g1aT<C> &B,
                                                        bool sgl) {
    using T = typename C::F;
    pair_case_t<C> r;
    if (sgl || g1a_is_inf<C>(B)) {
        r.kind = 1;
        r.copy = A;
        return r;
    }
    if (g1a_is_inf<C>(A)) {
        r.kind = 1;
        r.copy = B;
        return r;
    }
    if (fe9_eq_modp<T>(A.x, B.x)) {
        if (fe9_eq_modp<T>(A.y, B.y) && !fe9_is_zero_modp<T>(A.y)) {
            // doubling: lambda = 3 x1^2 / 2 y1
            r.kind = 0;
            r.d = add9_n<T>(A.y, A.y);
            feL<C::F::L> s = mont_sqr9<T>(A.x);
            r.n = add9_n<T>(add9_n<T>(s, s), s);
            return r;
        }
        r.kind = 1;  // B = -A (or 2-torsion): identity
        r.copy.x = fe9z<C::F::L>();
        r.copy.y = fe9z<C::F::L>();
        return r;
    }
    r.kind = 0;
    r.d = subn9<T>(B.x, A.x);
    r.n = subn9<T>(B.y, A.y);
    return r;
}

// largest b with off[b] <= j (off non-decreasing, off[0]=0)
__device__ __forceinline__ uint32_t bucket_of(const uint32_t *__restrict__ off,
                                              uint32_t nbuckets, uint32_t j) {
    uint32_t lo = 0, hi = nbuckets;
    while (lo < hi) {
        uint32_t mid = (lo + hi + 1) >> 1;
        if (off[mid] <= j)
            lo = mid;
        else
            hi = mid - 1;
    }
    return lo;
}

// one tree level.  L0 gathers pts[vals[.]] (the sorted-pair view of the
// input points); deeper levels read the previous level buffer directly.
//
// Pair assignment is LANE-STRIDED: a wave's 64 lanes cover 64 consecutive
// pair slots per step (lane l owns pairs base + i*64 + l), so level-buffer
// reads, SoA stores and out[] writes all coalesce — the per-thread
// contiguous-chunk version measured 44% of wave cycles parked on memory
// waits.  The forward pass classifies, chains the denominator prefix
// products and stores an SoA record (d, n, s=x1+x2, x1, y1) per lambda
// pair; copy-like pairs (identity operands, P+(-P), odd singleton) write
// out[] immediately and flag d=0.  The backward pass then needs NO point
// gathers and no re-classification: it peels inverses from the aux plane
// and finishes x3 = lam^2 - s, y3 = lam(x1-x3) - y1.
enum { AUX_PP = 0, AUX_D, AUX_N, AUX_S, AUX_X1, AUX_Y1, AUX_PLANES };

template <typename C, bool L0>
__global__ void __launch_bounds__(256)
k_pair_level(const g1aT<C> *__restrict__ pts, const uint32_t *__restrict__ vals,
             const uint32_t *__restrict__ off_in,
             const uint32_t *__restrict__ off_out,
             feL<C::F::L> *__restrict__ aux, uint32_t cap,
             g1aT<C> *__restrict__ out, uint32_t nbuckets) {
    using T = typename C::F;
    constexpr int LN = C::F::L;
    uint32_t total = off_out[nbuckets];
    uint32_t lane = threadIdx.x & 63u;
    uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    uint32_t base = wave * 64u * (uint32_t)PAIR_K;
    if (base >= total) return;
    feL<LN> *PPp = aux + (size_t)AUX_PP * cap;
    feL<LN> *Dp = aux + (size_t)AUX_D * cap;
    feL<LN> *Np = aux + (size_t)AUX_N * cap;
    feL<LN> *Sp = aux + (size_t)AUX_S * cap;
    feL<LN> *X1p = aux + (size_t)AUX_X1 * cap;
    feL<LN> *Y1p = aux + (size_t)AUX_Y1 * cap;
    uint32_t jf = base + lane;
    uint32_t bb = bucket_of(off_out, nbuckets, jf < total ? jf : total - 1);
    uint32_t ob = off_out[bb], oe = off_out[bb + 1];
    uint32_t ib = off_in[bb], ie = off_in[bb + 1];
    // forward: classify, store SoA records, chain prefix products
    feL<LN> PP = fe9_load<LN>(T::ONE);
    for (int i = 0; i < PAIR_K; i++) {
        uint32_t j = base + (uint32_t)i * 64u + lane;
        if (j >= total) break;
        while (j >= oe) {
            bb++;
            ob = oe;
            oe = off_out[bb + 1];
            ib = ie;
            ie = off_in[bb + 1];
        }
        uint32_t s0 = ib + 2 * (j - ob);
        bool sgl = s0 + 1 >= ie;
        g1aT<C> A = L0 ? pts[vals[s0]] : pts[s0];
        g1aT<C> B;
        if (!sgl) B = L0 ? pts[vals[s0 + 1]] : pts[s0 + 1];
        pair_case_t<C> pc = pair_classify<C>(A, B, sgl);
        if (pc.kind == 0) {
            PP = mont_mul9<T>(PP, pc.d);
            Dp[j] = pc.d;
            Np[j] = pc.n;
            Sp[j] = add9_n<T>(A.x, B.x);
            X1p[j] = A.x;
            Y1p[j] = A.y;
        } else {
            out[j] = pc.copy;
            Dp[j] = fe9z<LN>();  // d = 0 flags "done in forward"
        }
        PPp[j] = PP;
    }
    feL<LN> inv = mont_inv9<T>(PP);
    // backward: peel per-pair inverses from the SoA planes, emit sums
    for (int i = PAIR_K; i-- > 0;) {
        uint32_t j = base + (uint32_t)i * 64u + lane;
        if (j >= total) continue;
        feL<LN> d = Dp[j];
        if (fe9_is_zero_raw<LN>(d)) continue;
        feL<LN> invd = i > 0 ? mont_mul9<T>(inv, PPp[j - 64]) : inv;
        feL<LN> lam = mont_mul9<T>(Np[j], invd);
        feL<LN> x3 = subn9<T>(mont_sqr9<T>(lam), Sp[j]);
        feL<LN> y3 =
            subn9<T>(mont_mul9<T>(lam, subn9<T>(X1p[j], x3)), Y1p[j]);
        out[j].x = x3;
        out[j].y = y3;
        inv = mont_mul9<T>(inv, d);
    }
}

// ---- two-level running-sum reduction ----
// level 1: per (window, 32-bucket segment): from the top digit down,
//   run  += B_d           (=> run  = sum of segment buckets)
//   wsum += run           (=> wsum = sum (d - lo + 1) * B_d)
template <typename C, typename CFG>
__global__ void __launch_bounds__(256)
k_segment_reduce(const g1jT<C> *__restrict__ buckets,
                 g1jT<C> *__restrict__ seg_sum,
                 g1jT<C> *__restrict__ seg_wsum) {
    // the second accumulator lives in LDS: two register XYZZ accumulators
    // plus mul temporaries spill 232 B/lane to scratch otherwise
    __shared__ g1jT<C> lds_wsum[256];
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= CFG::NWIN * CFG::NSEG) return;
    uint32_t w = t / CFG::NSEG, seg = t % CFG::NSEG;
    uint32_t lo = seg * CFG::SEG;
    g1jT<C> run = g1_inf9<C>();
    lds_wsum[threadIdx.x] = g1_inf9<C>();
    for (int32_t d = (int32_t)lo + CFG::SEG - 1; d >= (int32_t)lo; d--) {
        // unsigned: the digit-0 bucket is unused — skip the add but keep the
        // wsum step so segment 0 carries the same (d - lo + 1) weights.
        // signed: every bucket is live (bucket id = |digit| - 1).
        if (CFG::SIGNED || d != 0)
            run = g1_add9(run, buckets[((uint32_t)w << CFG::IB) | (uint32_t)d]);
        lds_wsum[threadIdx.x] = g1_add9(lds_wsum[threadIdx.x], run);
    }
    seg_sum[t] = run;
    seg_wsum[t] = lds_wsum[threadIdx.x];
}

// level 2: fully parallel weighted combine + LDS tree reduction.
// Bucket id b carries digit value D(b) = b (unsigned) or b + 1 (signed), so
//   W_w = sum_j [ wsum_j + (j*SEG + delta - 1) * sum_j ],  delta = SIGNED,
// (unsigned j=0 term: -sum_0; signed j=0 term: wsum alone).
// The window factor 2^(C*w) is NOT folded here any more: windows leave this
// pipeline UNSCALED and the final Horner (C doublings per window) runs on
// the HOST at delivery time, overlapped with the next step's GPU work — the
// per-thread doubling chains used to be the latency tail of this kernel
// (catastrophically so for Fp2/G2: 248 serial doublings on one lane).
template <typename C, typename CFG>
__global__ void __launch_bounds__(CFG::RED_BLOCK)
k_weighted_reduce(const g1jT<C> *__restrict__ seg_sum,
                  const g1jT<C> *__restrict__ seg_wsum,
                  g1jT<C> *__restrict__ partials /* CFG::NPART */) {
    __shared__ g1jT<C> lds[CFG::RED_BLOCK];
    uint32_t t = blockIdx.x * CFG::RED_BLOCK + threadIdx.x;
    bool live = t < (uint32_t)(CFG::NWIN * CFG::NSEG);
    uint32_t j = t % CFG::NSEG;
    g1jT<C> val = g1_inf9<C>();
    if (live) {
        g1jT<C> ws = seg_wsum[t];
        g1jT<C> ss = seg_sum[t];
        if (j == 0 && !CFG::SIGNED) {
            // weight -1: subtract sum_0
            if (!g1_is_inf9(ss)) g1_neg_y9<C>(ss);
            val = g1_add9(ws, ss);
        } else if (j == 0) {
            val = ws;  // signed: weight 0
        } else {
            uint32_t weight = j * CFG::SEG + (CFG::SIGNED ? 0 : -1);  // < 2^C
            g1jT<C> acc = g1_inf9<C>();
            for (int b = CFG::C; b >= 0; b--) {
                acc = g1_dbl9(acc);
                if ((weight >> b) & 1) acc = g1_add9(acc, ss);
            }
            val = g1_add9(ws, acc);
        }
    }
    if constexpr (CFG::TREE) {
        // one window per block: LDS tree -> one partial per block
        lds[threadIdx.x] = val;
        __syncthreads();
        for (int s = CFG::RED_BLOCK / 2; s > 0; s >>= 1) {
            if (threadIdx.x < (uint32_t)s) {
                g1jT<C> o = lds[threadIdx.x + s];
                g1jT<C> m = g1_add9(lds[threadIdx.x], o);
                lds[threadIdx.x] = m;
            }
            __syncthreads();
        }
        if (threadIdx.x == 0) partials[blockIdx.x] = lds[0];
    } else {
        // small config: blocks span windows; write per-segment partials
        if (live) partials[t] = val;
    }
}

// level 3: 16 threads, 8 partials each -> per-window sums (pre-scaled)
// level 3: one 64-lane block per window (serial per-window sums cost up to
// ~1 ms when PER_WIN is large)
template <typename C, typename CFG>
__global__ void __launch_bounds__(64)
k_window_sum(const g1jT<C> *__restrict__ partials,
             g1jT<C> *__restrict__ windows) {
    constexpr int PER_WIN = CFG::NPART / CFG::NWIN;
    __shared__ g1jT<C> lds[64];
    uint32_t w = blockIdx.x;
    uint32_t t = threadIdx.x;
    g1jT<C> acc = g1_inf9<C>();
    for (uint32_t b = t; b < (uint32_t)PER_WIN; b += 64)
        acc = g1_add9(acc, partials[w * PER_WIN + b]);
    lds[t] = acc;
    __syncthreads();
    for (int sh = 32; sh > 0; sh >>= 1) {
        if (t < (uint32_t)sh) {
            g1jT<C> o = lds[t + sh];
            g1jT<C> m = g1_add9(lds[t], o);
            lds[t] = m;
        }
        __syncthreads();
    }
    if (t == 0) windows[w] = lds[0];
}

// ---- window emission ----
// Windows leave the GPU UNSCALED as NWIN Jacobian wire records; the host
// folds the 2^(C*w) factors with a Horner pass (C doublings + 1 add per
// window, ~0.2 ms for BN254) at delivery time, fully overlapped with the
// next pipelined step's GPU work.  This removes the per-thread doubling
// chains that were the fixed latency tail of the reduction (≥6x scaling
// target needs the 2^21-shard tail small; G2's chains cost 9.7 ms alone).
template <typename C, typename CFG>
__global__ void __launch_bounds__(64)
k_emit_windows(const g1jT<C> *__restrict__ windows, uint8_t *__restrict__ out) {
    uint32_t w = threadIdx.x;
    if (w >= (uint32_t)CFG::NWIN) return;
    g1_jac_be9<C>(out + (size_t)w * pt_bytes<C>::JAC, windows[w]);
}

// ---- single-op kernels (zisk-mirror ABI + Jacobian combine) ----

static __global__ void k_g1_add_single(const uint8_t *in /* 128 B: p1||p2 */,
                                uint8_t *out, uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    g1j9 acc = g1_inf9();
#pragma unroll
    for (int k = 0; k < 2; k++) {
        fe9 x = to_mont9(fe9_from_be(in + 64 * k));
        fe9 y = to_mont9(fe9_from_be(in + 64 * k + 32));
        if (fe9_is_zero_modp(x) && fe9_is_zero_modp(y)) continue;
        g1a9 p{x, y};
        if (!g1a9_on_curve(p)) {
            atomicOr(err, 1u);
            return;
        }
        acc = g1_add_affine9(acc, p);
    }
    g1_to_affine_be9(out, acc);
}

static __global__ void k_g1_mul_single(const uint8_t *in /* 96 B: point||scalar */,
                                uint8_t *out, uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    fe9 x = to_mont9(fe9_from_be(in));
    fe9 y = to_mont9(fe9_from_be(in + 32));
    if (fe9_is_zero_modp(x) && fe9_is_zero_modp(y)) {
        for (int j = 0; j < 8; j++) ((u64 *)out)[j] = 0;
        return;
    }
    g1a9 p{x, y};
    if (!g1a9_on_curve(p)) {
        atomicOr(err, 1u);
        return;
    }
    fe4 k = from_mont<Fr>(to_mont<Fr>(fe_from_be(in + 64)));
    if (fe_is_zero(k)) {
        for (int j = 0; j < 8; j++) ((u64 *)out)[j] = 0;
        return;
    }
    g1_to_affine_be9(out, g1_scalar_mul9(p, k.v));
}

// combine count Jacobian partials (96-B BE canonical each) -> affine
template <typename C>
static __global__ void k_g1_combine(const uint8_t *__restrict__ in, size_t count,
                             uint8_t *__restrict__ out) {
    using F = typename C::F;
    constexpr int NB = F::W64 * 8;
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    g1jT<C> acc = g1_inf9<C>();
    for (size_t i = 0; i < count; i++) {
        g1jT<C> t;
        if (!g1_jac_from_be9<C>(t, in + (size_t)pt_bytes<C>::JAC * i))
            continue;  // infinity
        acc = g1_add9(acc, t);
    }
    g1_to_affine_be9(out, acc);
}

// ---- fixed-base mode (blob-KZG: the 4096 setup points are FIXED across
// blobs, so precompute 2^(12w)*P_i once — the window dimension collapses
// into the point table and the per-blob MSM runs one 4096-bucket window;
// the same idea as c-kzg's KZG_PRECOMPUTE fixed-base tables) ----

// FB_C = 13 so the TOP window spans bits 247..255 (scalars < 2^255): its
// digits spread over ~2^8 buckets instead of piling 4096 entries into a
// handful (measured 7 ms of latency-bound straggler buckets at FB_C=12).
constexpr int FB_C = 13;
constexpr int FB_NWIN = 20;  // ceil(256 / 13), raw 256-bit scalars
using CfgFB = msm_cfg<FB_C, FB_C>;  // NWIN=1: single merged window space

// P_ext[w*n + i] = 2^(12w) * P_i (affine); inf entries follow the base flag
template <typename C>
static __global__ void k_fb_precompute(const g1aT<C> *__restrict__ pts,
                                const uint8_t *__restrict__ inf, size_t n,
                                g1aT<C> *__restrict__ ext) {
    size_t e = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= n * FB_NWIN) return;
    size_t i = e % n;
    int w = (int)(e / n);
    if (inf[i]) {
        ext[e] = pts[i];
        return;
    }
    g1jT<C> acc;
    acc.x = pts[i].x;
    acc.y = pts[i].y;
    acc.zz = fe9_load<C::F::L>(C::F::ONE);
    acc.zzz = fe9_load<C::F::L>(C::F::ONE);
    for (int d = 0; d < FB_C * w; d++) acc = g1_dbl9(acc);
    ext[e] = g1_to_affine9(acc);
}

// digits for fixed-base: entry e = (w, i); key = digit only (single window
// space), value = index into the precomputed table
static __global__ void k_fb_digits(const fe4 *__restrict__ scalars,
                            const uint8_t *__restrict__ inf,
                            uint32_t *__restrict__ keys,
                            uint32_t *__restrict__ vals, size_t n) {
    size_t e = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (e >= n * FB_NWIN) return;
    size_t i = e % n;
    int w = (int)(e / n);
    uint32_t d = inf[i] ? 0 : msm_digit<FB_C>(scalars[i], w);
    keys[e] = d;
    vals[e] = (uint32_t)e;
}

// ============================================================================
// BLS12-381 G1 specifics (bls_blst.rs EIP-2537
// semantics: 48-B BE canonical coords, (0,0) infinity, on-curve check,
// r-subgroup check for MSM inputs, scalars as FULL 256-bit integers).
// ============================================================================

// 96-byte BE affine -> Montgomery(2^406) g1aB + infinity flag.
// err bits: 1 = non-canonical/off-curve/subgroup (details via last_error)
static __global__ void k_bls_parse_points(const uint8_t *__restrict__ in,
                                   g1aB *__restrict__ pts,
                                   uint8_t *__restrict__ inf, size_t n,
                                   uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe14 xr = feT_from_be<FpB14T>(in + 96 * i);
    fe14 yr = feT_from_be<FpB14T>(in + 96 * i + 48);
    if (fe9_geq_raw<14>(xr, bn254::FPB_P) || fe9_geq_raw<14>(yr, bn254::FPB_P)) {
        atomicOr(err, 2u);  // non-canonical coordinate (EIP-2537 reject)
        return;
    }
    if (fe9_is_zero_raw<14>(xr) && fe9_is_zero_raw<14>(yr)) {
        inf[i] = 1;
        pts[i].x = fe9z<14>();
        pts[i].y = fe9z<14>();
        return;
    }
    g1aB p{to_mont9<FpB14T>(xr), to_mont9<FpB14T>(yr)};
    inf[i] = 0;
    if (!g1a9_on_curve(p)) {
        atomicOr(err, 1u);
        return;
    }
    // subgroup check: r*P == infinity (read_g1_subgroup, bls_blst.rs:215-222)
    g1jB t = g1_scalar_mul9(p, bn254::FRB_ORDER, 4);
    if (!g1_is_inf9(t)) atomicOr(err, 4u);
    pts[i] = p;
}

// P_i = (start+i+1)*G
static __global__ void k_bls_gen_points(g1aB *__restrict__ pts,
                                 uint8_t *__restrict__ inf, size_t n,
                                 uint64_t start) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    u64 k[4] = {start + i + 1, 0, 0, 0};
    g1aB g = g1_generator9<BlsG1>();
    g1jB acc = g1_inf9<BlsG1>();
    for (int b = 63; b >= 0; b--) {
        acc = g1_dbl9(acc);
        if ((k[0] >> b) & 1) acc = g1_add_affine9(acc, g);
    }
    pts[i] = g1_to_affine9(acc);
    inf[i] = 0;
}

static __global__ void k_bls_points_to_be(const g1aB *__restrict__ pts,
                                   const uint8_t *__restrict__ inf,
                                   uint8_t *__restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (inf[i]) {
        for (int j = 0; j < 12; j++) ((u64 *)(out + 96 * i))[j] = 0;
        return;
    }
    feT_to_be<FpB14T>(out + 96 * i, from_mont9<FpB14T>(pts[i].x));
    feT_to_be<FpB14T>(out + 96 * i + 48, from_mont9<FpB14T>(pts[i].y));
}

// scalars: raw 256-bit big-endian -> u64[4] LE words (NO reduction:
// blst SCALAR_BITS = 256); windows cover all 256 bits (16 x 16)
static __global__ void k_bls_parse_scalars(const uint8_t *__restrict__ in,
                                    fe4 *__restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const u64 *w = (const u64 *)(in + 32 * i);
    out[i].v[3] = __builtin_bswap64(w[0]);
    out[i].v[2] = __builtin_bswap64(w[1]);
    out[i].v[1] = __builtin_bswap64(w[2]);
    out[i].v[0] = __builtin_bswap64(w[3]);
}

// single ops (parity probes, bls_blst.rs g1_add / p1_mult semantics)
static __global__ void k_bls_g1_add_single(const uint8_t *in /* 192 B */, uint8_t *out,
                                    uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    g1jB acc = g1_inf9<BlsG1>();
#pragma unroll
    for (int k = 0; k < 2; k++) {
        fe14 xr = feT_from_be<FpB14T>(in + 96 * k);
        fe14 yr = feT_from_be<FpB14T>(in + 96 * k + 48);
        if (fe9_geq_raw<14>(xr, bn254::FPB_P) ||
            fe9_geq_raw<14>(yr, bn254::FPB_P)) {
            atomicOr(err, 2u);
            return;
        }
        if (fe9_is_zero_raw<14>(xr) && fe9_is_zero_raw<14>(yr)) continue;
        g1aB p{to_mont9<FpB14T>(xr), to_mont9<FpB14T>(yr)};
        if (!g1a9_on_curve(p)) {
            atomicOr(err, 1u);
            return;
        }
        acc = g1_add_affine9(acc, p);  // g1_add has no subgroup requirement
    }
    g1_to_affine_be9(out, acc);
}

static __global__ void k_bls_g1_mul_single(const uint8_t *in /* 96+32 B */,
                                    uint8_t *out, uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    fe14 xr = feT_from_be<FpB14T>(in);
    fe14 yr = feT_from_be<FpB14T>(in + 48);
    if (fe9_geq_raw<14>(xr, bn254::FPB_P) || fe9_geq_raw<14>(yr, bn254::FPB_P)) {
        atomicOr(err, 2u);
        return;
    }
    if (fe9_is_zero_raw<14>(xr) && fe9_is_zero_raw<14>(yr)) {
        for (int j = 0; j < 12; j++) ((u64 *)out)[j] = 0;
        return;
    }
    g1aB p{to_mont9<FpB14T>(xr), to_mont9<FpB14T>(yr)};
    if (!g1a9_on_curve(p)) {
        atomicOr(err, 1u);
        return;
    }
    // full 256-bit scalar (big-endian -> LE words), no reduction
    const u64 *w = (const u64 *)(in + 96);
    u64 k[4] = {__builtin_bswap64(w[3]), __builtin_bswap64(w[2]),
                __builtin_bswap64(w[1]), __builtin_bswap64(w[0])};
    g1_to_affine_be9(out, g1_scalar_mul9(p, k, 4));
}


// ---- BLS12-381 G2 input kernels (EIP-2537 192-byte points) ----
using g1aG2 = g1aT<BlsG2>;
using g1jG2 = g1jT<BlsG2>;

// parse + validate: canonical coords, (0,0,0,0) identity, on-curve, and
// r-subgroup (bls_blst.rs read_g2_subgroup) when check_subgroup is set
static __global__ void k_bls_g2_parse_points(const uint8_t *__restrict__ in,
                                      g1aG2 *__restrict__ pts,
                                      uint8_t *__restrict__ inf, size_t n,
                                      uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const uint8_t *b = in + 192 * i;
    fe14 c[4];
    bool allz = true;
#pragma unroll
    for (int k = 0; k < 4; k++) {
        c[k] = feT_from_be<FpB14T>(b + 48 * k);
        if (fe9_geq_raw<14>(c[k], bn254::FPB_P)) {
            atomicOr(err, 2u);
            return;
        }
        if (!fe9_is_zero_raw<14>(c[k])) allz = false;
    }
    if (allz) {
        inf[i] = 1;
        pts[i].x = fp2_zero();
        pts[i].y = fp2_zero();
        return;
    }
    g1aG2 p;
    p.x = {to_mont9<FpB14T>(c[0]), to_mont9<FpB14T>(c[1])};
    p.y = {to_mont9<FpB14T>(c[2]), to_mont9<FpB14T>(c[3])};
    inf[i] = 0;
    if (!g1a9_on_curve<BlsG2>(p)) {
        atomicOr(err, 1u);
        return;
    }
    g1jG2 t = g1_scalar_mul9(p, bn254::FRB_ORDER, 4);
    if (!g1_is_inf9<BlsG2>(t)) atomicOr(err, 4u);
    pts[i] = p;
}

// P_i = (start+i+1) * G2gen
static __global__ void k_bls_g2_gen_points(g1aG2 *__restrict__ pts,
                                    uint8_t *__restrict__ inf, size_t n,
                                    uint64_t start) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    u64 k[4] = {start + i + 1, 0, 0, 0};
    g1aG2 g = g1_generator9<BlsG2>();
    g1jG2 acc = g1_inf9<BlsG2>();
    for (int b = 63; b >= 0; b--) {
        acc = g1_dbl9<BlsG2>(acc);
        if ((k[0] >> b) & 1) acc = g1_add_affine9<BlsG2>(acc, g);
    }
    pts[i] = g1_to_affine9<BlsG2>(acc);
    inf[i] = 0;
}

static __global__ void k_bls_g2_points_to_be(const g1aG2 *__restrict__ pts,
                                      const uint8_t *__restrict__ inf,
                                      uint8_t *__restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (inf[i]) {
        for (int j = 0; j < 24; j++) ((u64 *)(out + 192 * i))[j] = 0;
        return;
    }
    fp2_to_be(out + 192 * i, pts[i].x);
    fp2_to_be(out + 192 * i + 96, pts[i].y);
}

// single ops (bls_blst.rs g2_add / p2_mult semantics; add: no subgroup check)
__device__ __forceinline__ int bls_g2_parse_one(const uint8_t *b, g1aG2 &p,
                                                bool &is_inf) {
    fe14 c[4];
    bool allz = true;
#pragma unroll
    for (int k = 0; k < 4; k++) {
        c[k] = feT_from_be<FpB14T>(b + 48 * k);
        if (fe9_geq_raw<14>(c[k], bn254::FPB_P)) return 2;
        if (!fe9_is_zero_raw<14>(c[k])) allz = false;
    }
    if (allz) {
        is_inf = true;
        return 0;
    }
    p.x = {to_mont9<FpB14T>(c[0]), to_mont9<FpB14T>(c[1])};
    p.y = {to_mont9<FpB14T>(c[2]), to_mont9<FpB14T>(c[3])};
    is_inf = false;
    if (!g1a9_on_curve<BlsG2>(p)) return 1;
    return 0;
}

static __global__ void k_bls_g2_add_single(const uint8_t *in /* 384 B */,
                                    uint8_t *out, uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    g1jG2 acc = g1_inf9<BlsG2>();
#pragma unroll
    for (int k = 0; k < 2; k++) {
        g1aG2 p;
        bool inf;
        int rc = bls_g2_parse_one(in + 192 * k, p, inf);
        if (rc) {
            atomicOr(err, rc == 2 ? 2u : 1u);
            return;
        }
        if (!inf) acc = g1_add_affine9<BlsG2>(acc, p);
    }
    g1_to_affine_be9<BlsG2>(out, acc);
}

static __global__ void k_bls_g2_mul_single(const uint8_t *in /* 192 + 32 B */,
                                    uint8_t *out, uint32_t *err) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    g1aG2 p;
    bool inf;
    int rc = bls_g2_parse_one(in, p, inf);
    if (rc) {
        atomicOr(err, rc == 2 ? 2u : 1u);
        return;
    }
    if (inf) {
        for (int j = 0; j < 24; j++) ((u64 *)out)[j] = 0;
        return;
    }
    u64 k[4];
    const u64 *w = (const u64 *)(in + 192);
#pragma unroll
    for (int j = 0; j < 4; j++) k[j] = __builtin_bswap64(w[3 - j]);
    g1jG2 r = g1_scalar_mul9(p, k, 4);
    g1_to_affine_be9<BlsG2>(out, r);
}

}  // namespace em
