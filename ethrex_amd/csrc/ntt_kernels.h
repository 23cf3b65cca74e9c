// ============================================================================
// Radix-2 NTT over BN254 Fr — gfx950 kernels, on the fe9 (9x29-bit) field
// core (carry-free v_mad_u64_u32 columns; see gpu_field9.h).
//
// Transform (same definition as the oracle, oracle/bn254_oracle.c):
//   forward:  A_j = sum_i a_i w^(ij) mod r,  w = W28^(2^(28-log2 n))
//   inverse:  a_i = n^-1 sum_j A_j w^(-ij)
// In/out: 32-byte big-endian canonical Fr elements, natural order.
// Device-resident data is fe9 Montgomery(2^261), 36 B per element.
//
// Two paths (selected in api.hip):
//   13 <= logn <= 24: four-step fused (3 tiled transposes + 2 LDS row-NTT
//   passes); otherwise: bit-reverse + logn radix-2 stage launches.
// Both produce identical integer results (same DFT), parity-pinned against
// the oracle.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "gpu_field9.h"

namespace em {

// ---- conversion / validation ----

// BE bytes -> fe9 Montgomery; flags err if elem >= r (canonical required)
__global__ void k_fr_from_be(const uint8_t *__restrict__ in,
                             fe9 *__restrict__ out, size_t n,
                             uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const u64 *w = (const u64 *)(in + 32 * i);
    u64 v[4] = {__builtin_bswap64(w[3]), __builtin_bswap64(w[2]),
                __builtin_bswap64(w[1]), __builtin_bswap64(w[0])};
    fe9 raw = fe9_from_u64x4(v);
    if (fe9_geq_raw(raw, bn254::FR9_P)) atomicOr(err, 1u);
    out[i] = to_mont9<Fr9T>(raw);
}

__global__ void k_fr_to_be(const fe9 *__restrict__ in, uint8_t *__restrict__ out,
                           size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe9 c = from_mont9<Fr9T>(in[i]);
    u64 w[4];
    fe9_to_u64x4(w, c);
    u64 *o = (u64 *)(out + 32 * i);
    o[0] = __builtin_bswap64(w[3]);
    o[1] = __builtin_bswap64(w[2]);
    o[2] = __builtin_bswap64(w[1]);
    o[3] = __builtin_bswap64(w[0]);
}

// ---- twiddle generation: tw[j] = w^j (fe9 Montgomery), j in [0, count) ----
// w2k[k] = w^(2^k) precomputed on host.
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

// ---- bit-reverse permutation (in-place swap) ----
__global__ void k_bit_reverse(fe9 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    size_t j = __brevll(i) >> (64 - logn);
    if (j > i) {
        fe9 t = a[i];
        a[i] = a[j];
        a[j] = t;
    }
}

// ---- one radix-2 DIT stage (fallback path) ----
__global__ void k_ntt_stage(fe9 *__restrict__ a, const fe9 *__restrict__ tw,
                            size_t n, int logn, int s) {
    size_t t = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= (n >> 1)) return;
    size_t half = (size_t)1 << (s - 1);
    size_t j = t & (half - 1);
    size_t g = t >> (s - 1);
    size_t idx = (g << s) + j;
    fe9 u = a[idx];
    fe9 v = mont_mul9<Fr9T>(a[idx + half], tw[j << (logn - s)]);
    a[idx] = add9_n<Fr9T>(u, v);
    a[idx + half] = subm9<Fr9T>(u, v);  // b = mul output
}

// ---- four-step fused path (13 <= logn <= 24) ----
// n = N1*N2; A[r][c] = a[r*N2+c]:
//   T0: A1[c][r] = A[r][c]; P1: row NTT_N1 + w^(k1*c) twiddle;
//   T1; P2: row NTT_N2 (+ 1/n for iNTT); T2 -> natural order.

// tiled fe9 transpose, 32x32 tiles (+1 pad column for LDS banking);
// grid.z = batch of independent R x C sub-matrices at stride R*C
__global__ void __launch_bounds__(256)
k_transpose_fe9(const fe9 *__restrict__ src, fe9 *__restrict__ dst,
                uint32_t R, uint32_t C) {
    __shared__ fe9 tile[32][33];
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
__global__ void __launch_bounds__(1024)
k_ntt_row(fe9 *__restrict__ data, int logM, const fe9 *__restrict__ tw_row,
          const fe9 *__restrict__ tw_full, const fe9 *__restrict__ scale,
          uint32_t cmask = 0xffffffffu) {
    __shared__ fe9 smem[4096];
    const uint32_t M = 1u << logM;
    fe9 *row = data + (size_t)blockIdx.x * M;
    for (uint32_t i = threadIdx.x; i < M; i += blockDim.x) {
        uint32_t j = __brev(i) >> (32 - logM);
        smem[j] = row[i];
    }
    __syncthreads();
    // radix-2^2: two DIT stages per LDS round trip (same mul count as
    // radix-2 — 1 mul/element per 2 stages — but half the LDS traffic and
    // half the __syncthreads).  W2b = W2a * w^(M/4).
    int s = 1;
    for (; s + 1 <= logM; s += 2) {
        uint32_t q = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 2); t += blockDim.x) {
            uint32_t j = t & (q - 1);
            uint32_t idx = ((t >> (s - 1)) << (s + 1)) + j;
            fe9 w1 = tw_row[j << (logM - s)];
            fe9 a = smem[idx];
            fe9 b = mont_mul9<Fr9T>(smem[idx + q], w1);
            fe9 c = smem[idx + 2 * q];
            fe9 d = mont_mul9<Fr9T>(smem[idx + 3 * q], w1);
            fe9 t0 = add9_n<Fr9T>(a, b);
            fe9 t1 = subm9<Fr9T>(a, b);
            fe9 t2 = add9_n<Fr9T>(c, d);
            fe9 t3 = subm9<Fr9T>(c, d);
            fe9 u2 = mont_mul9<Fr9T>(t2, tw_row[j << (logM - s - 1)]);
            fe9 u3 = mont_mul9<Fr9T>(
                t3, tw_row[(j + q) << (logM - s - 1)]);
            smem[idx] = add9_n<Fr9T>(t0, u2);
            smem[idx + 2 * q] = subm9<Fr9T>(t0, u2);
            smem[idx + q] = add9_n<Fr9T>(t1, u3);
            smem[idx + 3 * q] = subm9<Fr9T>(t1, u3);
        }
        __syncthreads();
    }
    for (; s <= logM; s++) {  // odd-logM tail: one radix-2 stage
        uint32_t half = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 1); t += blockDim.x) {
            uint32_t j = t & (half - 1);
            uint32_t idx = ((t >> (s - 1)) << s) + j;
            fe9 u = smem[idx];
            fe9 v = mont_mul9<Fr9T>(smem[idx + half], tw_row[j << (logM - s)]);
            smem[idx] = add9_n<Fr9T>(u, v);
            smem[idx + half] = subm9<Fr9T>(u, v);
        }
        __syncthreads();
    }
    uint64_t c = blockIdx.x & cmask;
    for (uint32_t k = threadIdx.x; k < M; k += blockDim.x) {
        fe9 x = smem[k];
        if (tw_full) x = mont_mul9<Fr9T>(x, tw_full[(size_t)k * c]);
        if (scale) x = mont_mul9<Fr9T>(x, *scale);
        row[k] = x;
    }
}

// packed small-row NTT: 1024/M rows of length M <= 512 per 36-KB block
// (the 144-KB k_ntt_row runs one block/CU and idles 1-(M/4096) of its
// threads on the two-level path's 64/128-point inner rows).  One butterfly
// per thread per stage; rows packed smem[rr*M + j].
__global__ void __launch_bounds__(512)
k_ntt_row_small(fe9 *__restrict__ data, int logM,
                const fe9 *__restrict__ tw_row,
                const fe9 *__restrict__ tw_full,
                const fe9 *__restrict__ scale, uint32_t cmask) {
    __shared__ fe9 smem[1024];
    const uint32_t M = 1u << logM;
    const uint32_t RPB = 1024u >> logM;  // rows per block
    size_t row0 = (size_t)blockIdx.x * RPB;
    fe9 *base = data + row0 * M;
    for (uint32_t i = threadIdx.x; i < 1024; i += blockDim.x) {
        uint32_t rr = i >> logM, j = i & (M - 1);
        smem[rr * M + (__brev(j) >> (32 - logM))] = base[i];
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
        if (tw_full) x = mont_mul9<Fr9T>(x, tw_full[(size_t)kk * c]);
        if (scale) x = mont_mul9<Fr9T>(x, *scale);
        base[i] = x;
    }
}

// ---- scale by n^-1 (iNTT, fallback path) ----
__global__ void k_ntt_scale(fe9 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    a[i] = mont_mul9<Fr9T>(a[i], fe9_load(bn254::FR9_INV_POW2[logn]));
}

}  // namespace em
