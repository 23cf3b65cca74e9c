// ============================================================================
// Radix-2 NTT over BN254 Fr — gfx950 kernels, on the fe9 (9x29-bit) field
// core (carry-free v_mad_u64_u32 columns; see gpu_field9.h).
//
// Transform (same definition as the oracle, oracle/bn254_oracle.c):
//   forward:  A_j = sum_i a_i w^(ij) mod r,  w = W28^(2^(28-log2 n))
//   inverse:  a_i = n^-1 sum_j A_j w^(-ij)
// In/out: 32-byte big-endian canonical Fr elements, natural order.
//
// RESIDENT FORMAT (round 2): the HBM-resident vector and the large twiddle
// tables are PACKED 4x64 Montgomery values ("fe4m", 32 B/element, value
// < 2p < 2^255) — the transform is HBM-bound and the 36-B fe9 format cost
// ~11% extra traffic plus line-straddling gathers.  Elements are unpacked
// to fe9 on load (pure bit repacking, ~30 VALU ops) and repacked on store;
// all arithmetic stays on the 29-bit-limb core.  Small per-row twiddle
// tables (<= 72 KB, L2-resident) stay fe9 to keep per-butterfly loads
// conversion-free.
//
// P1's inter-step twiddle w^(k*c) is read from a REORDERED table
// TW2[c][k] = w^(k*c) (c = row, k = element) so the per-row access is a
// coalesced stream instead of the old stride-c gather over an n-sized
// table (which fetched a full cache line per element).
//
// Two row-kernel shapes, A/B-measured (profiles/r02_summary.md):
// radix-2^2 at 1024 threads (4 waves/SIMD; the DEFAULT — occupancy hides
// the LDS+mul latency best) and radix-2^3 at 512 threads (3 DIT stages
// per LDS round trip, 1/3 the LDS traffic; EM_NTT_R8 selects it).  Both
// use a SKEWED LDS layout — the bit-reversed load scatter otherwise hits
// one bank group wave-wide — and dynamic LDS so 2048-element rows run 2
// blocks/CU.
//
// Two paths (selected in api_ntt.hip):
//   13 <= logn <= 24: four-step fused (3 tiled transposes + 2 LDS row-NTT
//   passes); 25-26: two-level four-step; otherwise bit-reverse + logn
//   radix-2 stage launches.  All produce identical integer results (same
//   DFT), parity-pinned against the oracle.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "gpu_field.h"   // fe4: the packed 4x64 resident format
#include "gpu_field9.h"

namespace em {

// ---- packed 4x64 Montgomery <-> fe9 (values norm2p < 2p < 2^255) ----
__device__ __forceinline__ fe9 fe4m_unpack(const fe4 &x) {
    return fe9_from_u64x4(x.v);
}
__device__ __forceinline__ fe4 fe4m_pack(const fe9 &x) {
    fe4 r;
    fe9_to_u64x4(r.v, x);
    return r;
}

// ---- conversion / validation ----

// BE bytes -> packed Montgomery; flags err if elem >= r (canonical required)
__global__ void k_fr_from_be(const uint8_t *__restrict__ in,
                             fe4 *__restrict__ out, size_t n,
                             uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const u64 *w = (const u64 *)(in + 32 * i);
    u64 v[4] = {__builtin_bswap64(w[3]), __builtin_bswap64(w[2]),
                __builtin_bswap64(w[1]), __builtin_bswap64(w[0])};
    fe9 raw = fe9_from_u64x4(v);
    if (fe9_geq_raw(raw, bn254::FR9_P)) atomicOr(err, 1u);
    out[i] = fe4m_pack(to_mont9<Fr9T>(raw));
}

__global__ void k_fr_to_be(const fe4 *__restrict__ in, uint8_t *__restrict__ out,
                           size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe9 c = from_mont9<Fr9T>(fe4m_unpack(in[i]));
    u64 w[4];
    fe9_to_u64x4(w, c);
    u64 *o = (u64 *)(out + 32 * i);
    o[0] = __builtin_bswap64(w[3]);
    o[1] = __builtin_bswap64(w[2]);
    o[2] = __builtin_bswap64(w[1]);
    o[3] = __builtin_bswap64(w[0]);
}

// ---- twiddle generation ----
// row tables (fe9, small): tw[j] = w^j, j in [0, count); w2k[k] = w^(2^k).
__global__ void k_gen_twiddles(fe9 *__restrict__ tw, size_t count,
                               const fe9 *__restrict__ w2k, int bits) {
    size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= count) return;
    fe9 acc = fe9_load(bn254::FR9_ONE);
    size_t e = j;
    for (int k = 0; k < bits && e; k++, e >>= 1)
        if (e & 1) acc = mont_mul9<Fr9T>(acc, w2k[k]);
    tw[j] = acc;
}

// reordered P1 table (fe4m, streamed): TW2[c*M + k] = w^(k*c), c < rows,
// k < M.  exponent k*c < M*rows = transform size, so <= `bits` bits.
__global__ void k_gen_tw2(fe4 *__restrict__ tw, uint32_t M, size_t rows,
                          const fe9 *__restrict__ w2k, int bits) {
    size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= (size_t)M * rows) return;
    size_t c = j / M, k = j % M;
    fe9 acc = fe9_load(bn254::FR9_ONE);
    u64 e = (u64)c * k;
    for (int b = 0; b < bits && e; b++, e >>= 1)
        if (e & 1) acc = mont_mul9<Fr9T>(acc, w2k[b]);
    tw[j] = fe4m_pack(acc);
}

// ---- bit-reverse permutation (in-place swap; fallback path) ----
__global__ void k_bit_reverse(fe4 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    size_t j = __brevll(i) >> (64 - logn);
    if (j > i) {
        fe4 t = a[i];
        a[i] = a[j];
        a[j] = t;
    }
}

// ---- one radix-2 DIT stage (fallback path; tw table fe9) ----
__global__ void k_ntt_stage(fe4 *__restrict__ a, const fe9 *__restrict__ tw,
                            size_t n, int logn, int s) {
    size_t t = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= (n >> 1)) return;
    size_t half = (size_t)1 << (s - 1);
    size_t j = t & (half - 1);
    size_t g = t >> (s - 1);
    size_t idx = (g << s) + j;
    fe9 u = fe4m_unpack(a[idx]);
    fe9 v = mont_mul9<Fr9T>(fe4m_unpack(a[idx + half]), tw[j << (logn - s)]);
    a[idx] = fe4m_pack(add9_n<Fr9T>(u, v));
    a[idx + half] = fe4m_pack(subm9<Fr9T>(u, v));  // b = mul output
}

// ---- four-step fused path ----
// n = N1*N2; A[r][c] = a[r*N2+c]:
//   T0: A1[c][r] = A[r][c]; P1: row NTT_N1 + TW2 twiddle;
//   T1; P2: row NTT_N2 (+ 1/n for iNTT); T2 -> natural order.

// tiled fe4 transpose, 32x32 tiles (+1 pad column for LDS banking);
// grid.z = batch of independent R x C sub-matrices at stride R*C
__global__ void __launch_bounds__(256)
k_transpose_fe4(const fe4 *__restrict__ src, fe4 *__restrict__ dst,
                uint32_t R, uint32_t C) {
    __shared__ fe4 tile[32][33];
    size_t base = (size_t)blockIdx.z * R * C;
    uint32_t c0 = blockIdx.x * 32, r0 = blockIdx.y * 32;
    uint32_t tx = threadIdx.x & 31, ty = threadIdx.x >> 5;  // 8 rows/pass
    for (uint32_t dy = ty; dy < 32; dy += 8)
        tile[dy][tx] = src[base + (size_t)(r0 + dy) * C + c0 + tx];
    __syncthreads();
    for (uint32_t dy = ty; dy < 32; dy += 8)
        dst[base + (size_t)(c0 + dy) * R + r0 + tx] = tile[tx][dy];
}

// one row NTT of length M = 2^logM fully in LDS (fe9: 4096*36 B = 144 KiB).
// 512 threads x 8 elements; radix-2^3 rounds (three DIT stages in
// registers per LDS round trip), radix-2^2 / radix-2 tail for logM % 3.
// tw2 (fe4m, may be null): output k scaled by TW2[c*M + k], c = blockIdx
// & cmask; scale (fe9, may be null): iNTT 1/n factor.
// skewed LDS indexing: physical slot = i + (i >> 6).  The bit-reversed
// load scatter otherwise lands a whole wave on one bank group (rev of
// consecutive i strides 2048 elements; 9*2048 dwords = 0 mod 64 banks
// = a 64-way conflict); with the skew, lane l's slot moves by
// rev6(l)*65 -> 9*rev6(l) mod 64, a permutation of the banks.
#define EM_SK(i) ((i) + ((i) >> 6))
__global__ void __launch_bounds__(512)
k_ntt_row(fe4 *__restrict__ data, int logM, const fe9 *__restrict__ tw_row,
          const fe4 *__restrict__ tw2, const fe9 *__restrict__ scale,
          uint32_t cmask = 0xffffffffu) {
    // dynamic LDS, (M + M/64) fe9 slots: 2048-element rows then run 2
    // blocks/CU (73 KiB) instead of being pinned at the 4096-row footprint
    extern __shared__ fe9 smem[];
    const uint32_t M = 1u << logM;
    fe4 *row = data + (size_t)blockIdx.x * M;
    for (uint32_t i = threadIdx.x; i < M; i += blockDim.x) {
        uint32_t j = __brev(i) >> (32 - logM);
        smem[EM_SK(j)] = fe4m_unpack(row[i]);
    }
    __syncthreads();
    int s = 1;
    // radix-2^3 rounds: stages s, s+1, s+2 on 8 register-resident elements
    for (; s + 2 <= logM; s += 3) {
        uint32_t q = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 3); t += blockDim.x) {
            uint32_t j = t & (q - 1);
            uint32_t base = ((t >> (s - 1)) << (s + 2)) + j;
            fe9 e0 = smem[EM_SK(base)];
            fe9 e1 = smem[EM_SK(base + q)];
            fe9 e2 = smem[EM_SK(base + 2 * q)];
            fe9 e3 = smem[EM_SK(base + 3 * q)];
            fe9 e4 = smem[EM_SK(base + 4 * q)];
            fe9 e5 = smem[EM_SK(base + 5 * q)];
            fe9 e6 = smem[EM_SK(base + 6 * q)];
            fe9 e7 = smem[EM_SK(base + 7 * q)];
            // stage s: stride-q pairs, shared twiddle
            fe9 w0 = tw_row[j << (logM - s)];
            fe9 v;
            v = mont_mul9<Fr9T>(e1, w0);
            e1 = subm9<Fr9T>(e0, v);
            e0 = add9_n<Fr9T>(e0, v);
            v = mont_mul9<Fr9T>(e3, w0);
            e3 = subm9<Fr9T>(e2, v);
            e2 = add9_n<Fr9T>(e2, v);
            v = mont_mul9<Fr9T>(e5, w0);
            e5 = subm9<Fr9T>(e4, v);
            e4 = add9_n<Fr9T>(e4, v);
            v = mont_mul9<Fr9T>(e7, w0);
            e7 = subm9<Fr9T>(e6, v);
            e6 = add9_n<Fr9T>(e6, v);
            // stage s+1: stride-2q pairs, twiddles at j and j+q
            fe9 wa = tw_row[j << (logM - s - 1)];
            fe9 wb = tw_row[(j + q) << (logM - s - 1)];
            v = mont_mul9<Fr9T>(e2, wa);
            e2 = subm9<Fr9T>(e0, v);
            e0 = add9_n<Fr9T>(e0, v);
            v = mont_mul9<Fr9T>(e3, wb);
            e3 = subm9<Fr9T>(e1, v);
            e1 = add9_n<Fr9T>(e1, v);
            v = mont_mul9<Fr9T>(e6, wa);
            e6 = subm9<Fr9T>(e4, v);
            e4 = add9_n<Fr9T>(e4, v);
            v = mont_mul9<Fr9T>(e7, wb);
            e7 = subm9<Fr9T>(e5, v);
            e5 = add9_n<Fr9T>(e5, v);
            // stage s+2: stride-4q pairs, twiddles at j, j+q, j+2q, j+3q
            fe9 w4a = tw_row[j << (logM - s - 2)];
            fe9 w4b = tw_row[(j + q) << (logM - s - 2)];
            fe9 w4c = tw_row[(j + 2 * q) << (logM - s - 2)];
            fe9 w4d = tw_row[(j + 3 * q) << (logM - s - 2)];
            v = mont_mul9<Fr9T>(e4, w4a);
            e4 = subm9<Fr9T>(e0, v);
            e0 = add9_n<Fr9T>(e0, v);
            v = mont_mul9<Fr9T>(e5, w4b);
            e5 = subm9<Fr9T>(e1, v);
            e1 = add9_n<Fr9T>(e1, v);
            v = mont_mul9<Fr9T>(e6, w4c);
            e6 = subm9<Fr9T>(e2, v);
            e2 = add9_n<Fr9T>(e2, v);
            v = mont_mul9<Fr9T>(e7, w4d);
            e7 = subm9<Fr9T>(e3, v);
            e3 = add9_n<Fr9T>(e3, v);
            smem[EM_SK(base)] = e0;
            smem[EM_SK(base + q)] = e1;
            smem[EM_SK(base + 2 * q)] = e2;
            smem[EM_SK(base + 3 * q)] = e3;
            smem[EM_SK(base + 4 * q)] = e4;
            smem[EM_SK(base + 5 * q)] = e5;
            smem[EM_SK(base + 6 * q)] = e6;
            smem[EM_SK(base + 7 * q)] = e7;
        }
        __syncthreads();
    }
    // radix-2^2 tail (logM % 3 == 2)
    for (; s + 1 <= logM; s += 2) {
        uint32_t q = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 2); t += blockDim.x) {
            uint32_t j = t & (q - 1);
            uint32_t idx = ((t >> (s - 1)) << (s + 1)) + j;
            fe9 w1 = tw_row[j << (logM - s)];
            fe9 a = smem[EM_SK(idx)];
            fe9 b = mont_mul9<Fr9T>(smem[EM_SK(idx + q)], w1);
            fe9 c = smem[EM_SK(idx + 2 * q)];
            fe9 d = mont_mul9<Fr9T>(smem[EM_SK(idx + 3 * q)], w1);
            fe9 t0 = add9_n<Fr9T>(a, b);
            fe9 t1 = subm9<Fr9T>(a, b);
            fe9 t2 = add9_n<Fr9T>(c, d);
            fe9 t3 = subm9<Fr9T>(c, d);
            fe9 u2 = mont_mul9<Fr9T>(t2, tw_row[j << (logM - s - 1)]);
            fe9 u3 = mont_mul9<Fr9T>(t3, tw_row[(j + q) << (logM - s - 1)]);
            smem[EM_SK(idx)] = add9_n<Fr9T>(t0, u2);
            smem[EM_SK(idx + 2 * q)] = subm9<Fr9T>(t0, u2);
            smem[EM_SK(idx + q)] = add9_n<Fr9T>(t1, u3);
            smem[EM_SK(idx + 3 * q)] = subm9<Fr9T>(t1, u3);
        }
        __syncthreads();
    }
    // radix-2 tail (logM % 3 == 1)
    for (; s <= logM; s++) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 1); t += blockDim.x) {
            uint32_t j = t & (half - 1);
            uint32_t idx = ((t >> (s - 1)) << s) + j;
            fe9 u = smem[EM_SK(idx)];
            fe9 v = mont_mul9<Fr9T>(smem[EM_SK(idx + half)], tw_row[j << (logM - s)]);
            smem[EM_SK(idx)] = add9_n<Fr9T>(u, v);
            smem[EM_SK(idx + half)] = subm9<Fr9T>(u, v);
        }
        __syncthreads();
    }
    uint64_t c = blockIdx.x & cmask;
    const fe4 *t2row = tw2 ? tw2 + (size_t)c * M : nullptr;
    for (uint32_t k = threadIdx.x; k < M; k += blockDim.x) {
        fe9 x = smem[EM_SK(k)];
        if (t2row) x = mont_mul9<Fr9T>(x, fe4m_unpack(t2row[k]));
        if (scale) x = mont_mul9<Fr9T>(x, *scale);
        row[k] = fe4m_pack(x);
    }
}


// radix-2^2 row variant (A/B occupancy experiment, EM_NTT_R4): 4 elements
// per thread per round trip, launchable at 512/768/1024 threads — more
// waves/SIMD than the radix-2^3 shape at the cost of 1.5x the LDS round
// trips.  Same skewed layout and IO as k_ntt_row.
__global__ void __launch_bounds__(1024)
k_ntt_row4(fe4 *__restrict__ data, int logM, const fe9 *__restrict__ tw_row,
           const fe4 *__restrict__ tw2, const fe9 *__restrict__ scale,
           uint32_t cmask) {
    extern __shared__ fe9 smem[];
    const uint32_t M = 1u << logM;
    fe4 *row = data + (size_t)blockIdx.x * M;
    for (uint32_t i = threadIdx.x; i < M; i += blockDim.x) {
        uint32_t j = __brev(i) >> (32 - logM);
        smem[EM_SK(j)] = fe4m_unpack(row[i]);
    }
    __syncthreads();
    int s = 1;
    for (; s + 1 <= logM; s += 2) {
        uint32_t q = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 2); t += blockDim.x) {
            uint32_t j = t & (q - 1);
            uint32_t idx = ((t >> (s - 1)) << (s + 1)) + j;
            fe9 w1 = tw_row[j << (logM - s)];
            fe9 a = smem[EM_SK(idx)];
            fe9 b = mont_mul9<Fr9T>(smem[EM_SK(idx + q)], w1);
            fe9 c = smem[EM_SK(idx + 2 * q)];
            fe9 d = mont_mul9<Fr9T>(smem[EM_SK(idx + 3 * q)], w1);
            fe9 t0 = add9_n<Fr9T>(a, b);
            fe9 t1 = subm9<Fr9T>(a, b);
            fe9 t2 = add9_n<Fr9T>(c, d);
            fe9 t3 = subm9<Fr9T>(c, d);
            fe9 u2 = mont_mul9<Fr9T>(t2, tw_row[j << (logM - s - 1)]);
            fe9 u3 = mont_mul9<Fr9T>(t3, tw_row[(j + q) << (logM - s - 1)]);
            smem[EM_SK(idx)] = add9_n<Fr9T>(t0, u2);
            smem[EM_SK(idx + 2 * q)] = subm9<Fr9T>(t0, u2);
            smem[EM_SK(idx + q)] = add9_n<Fr9T>(t1, u3);
            smem[EM_SK(idx + 3 * q)] = subm9<Fr9T>(t1, u3);
        }
        __syncthreads();
    }
    for (; s <= logM; s++) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 1); t += blockDim.x) {
            uint32_t j = t & (half - 1);
            uint32_t idx = ((t >> (s - 1)) << s) + j;
            fe9 u = smem[EM_SK(idx)];
            fe9 v = mont_mul9<Fr9T>(smem[EM_SK(idx + half)],
                                    tw_row[j << (logM - s)]);
            smem[EM_SK(idx)] = add9_n<Fr9T>(u, v);
            smem[EM_SK(idx + half)] = subm9<Fr9T>(u, v);
        }
        __syncthreads();
    }
    uint64_t c = blockIdx.x & cmask;
    const fe4 *t2row = tw2 ? tw2 + (size_t)c * M : nullptr;
    for (uint32_t k = threadIdx.x; k < M; k += blockDim.x) {
        fe9 x = smem[EM_SK(k)];
        if (t2row) x = mont_mul9<Fr9T>(x, fe4m_unpack(t2row[k]));
        if (scale) x = mont_mul9<Fr9T>(x, *scale);
        row[k] = fe4m_pack(x);
    }
}

// packed small-row NTT: 1024/M rows of length M <= 512 per block (the
// 144-KB k_ntt_row runs one block/CU and would idle 1-(M/4096) of its
// threads on the two-level path's 64/128-point inner rows).  One butterfly
// per thread per stage; rows packed smem[rr*M + j].  tw2 is the reordered
// [c][k] table of the INNER transform (c = row index & cmask).
__global__ void __launch_bounds__(512)
k_ntt_row_small(fe4 *__restrict__ data, int logM,
                const fe9 *__restrict__ tw_row,
                const fe4 *__restrict__ tw2,
                const fe9 *__restrict__ scale, uint32_t cmask) {
    __shared__ fe9 smem[1024];
    const uint32_t M = 1u << logM;
    size_t row0 = (size_t)blockIdx.x * (1024u >> logM);
    fe4 *base = data + row0 * M;
    for (uint32_t i = threadIdx.x; i < 1024; i += blockDim.x) {
        uint32_t rr = i >> logM, j = i & (M - 1);
        smem[rr * M + (__brev(j) >> (32 - logM))] = fe4m_unpack(base[i]);
    }
    __syncthreads();
    for (int s = 1; s <= logM; s++) {
        uint32_t half = 1u << (s - 1);
        uint32_t t = threadIdx.x;  // 512 butterflies = 1024 elements
        uint32_t rr = t >> (logM - 1), ti = t & ((M >> 1) - 1);
        uint32_t j = ti & (half - 1);
        uint32_t idx = rr * M + ((ti >> (s - 1)) << s) + j;
        fe9 u = smem[idx];
        fe9 v = mont_mul9<Fr9T>(smem[idx + half], tw_row[j << (logM - s)]);
        smem[idx] = add9_n<Fr9T>(u, v);
        smem[idx + half] = subm9<Fr9T>(u, v);
        __syncthreads();
    }
    for (uint32_t i = threadIdx.x; i < 1024; i += blockDim.x) {
        uint32_t rr = i >> logM, kk = i & (M - 1);
        uint64_t c = (uint64_t)((row0 + rr) & cmask);
        fe9 x = smem[i];
        if (tw2) x = mont_mul9<Fr9T>(x, fe4m_unpack(tw2[(size_t)c * M + kk]));
        if (scale) x = mont_mul9<Fr9T>(x, *scale);
        base[i] = fe4m_pack(x);
    }
}

// ---- scale by n^-1 (iNTT, fallback path) ----
__global__ void k_ntt_scale(fe4 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    a[i] = fe4m_pack(mont_mul9<Fr9T>(fe4m_unpack(a[i]),
                                     fe9_load(bn254::FR9_INV_POW2[logn])));
}

}  // namespace em
