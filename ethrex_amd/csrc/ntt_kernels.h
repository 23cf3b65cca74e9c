// ============================================================================
// Radix-2 NTT over BN254 Fr — gfx950 kernels.
//
// Transform (same definition as the oracle, oracle/bn254_oracle.c):
//   forward:  A_j = sum_i a_i w^(ij) mod r,  w = W28^(2^(28-log2 n))
//   inverse:  a_i = n^-1 sum_j A_j w^(-ij)
// In/out: 32-byte big-endian canonical Fr elements, natural order.
//
// HBM-bound path: data stays resident in fe4 (Montgomery) form; one stage
// kernel pass streams 2 x 32 B per butterfly pair.  Twiddle tables
// (n/2 entries, forward and inverse) are built once per plan on device.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "gpu_field.h"

namespace em {

// ---- conversion / validation ----

// BE bytes -> Montgomery fe4; flags err if elem >= r (canonical required)
__global__ void k_fr_from_be(const uint8_t *__restrict__ in,
                             fe4 *__restrict__ out, size_t n,
                             uint32_t *__restrict__ err) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe4 v = fe_from_be(in + 32 * i);
    if (fe_geq(v, fe4{{Fr::MOD[0], Fr::MOD[1], Fr::MOD[2], Fr::MOD[3]}}))
        atomicOr(err, 1u);
    out[i] = to_mont<Fr>(v);
}

__global__ void k_fr_to_be(const fe4 *__restrict__ in, uint8_t *__restrict__ out,
                           size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe_to_be(out + 32 * i, from_mont<Fr>(in[i]));
}

// ---- twiddle generation: tw[j] = w^j (Montgomery), j in [0, n/2) ----
// w2k[k] = w^(2^k) precomputed on host (gpu_field host path).
__global__ void k_gen_twiddles(fe4 *__restrict__ tw, size_t half,
                               const fe4 *__restrict__ w2k, int logn) {
    size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= half) return;
    fe4 acc = fe_one_mont<Fr>();
    size_t e = j;
    for (int k = 0; k < logn && e; k++, e >>= 1)
        if (e & 1) acc = mont_mul<Fr>(acc, w2k[k]);
    tw[j] = acc;
}

// ---- bit-reverse permutation (in-place swap) ----
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

// ---- one radix-2 DIT stage ----
// stage s (1-based): m = 2^s, half = m/2; thread t handles butterfly
// (g*m + j, g*m + j + half), twiddle tw[j << (logn - s)].
__global__ void k_ntt_stage(fe4 *__restrict__ a, const fe4 *__restrict__ tw,
                            size_t n, int logn, int s) {
    size_t t = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= (n >> 1)) return;
    size_t half = (size_t)1 << (s - 1);
    size_t j = t & (half - 1);
    size_t g = t >> (s - 1);
    size_t idx = (g << s) + j;
    fe4 u = a[idx];
    fe4 v = mont_mul<Fr>(a[idx + half], tw[j << (logn - s)]);
    a[idx] = mod_add<Fr>(u, v);
    a[idx + half] = mod_sub<Fr>(u, v);
}

// ---- four-step fused path (13 <= logn <= 24) ----
// n = N1*N2; A[r][c] = a[r*N2+c].  Five passes instead of logn:
//   T0: A1[c][r] = A[r][c]            (tiled transpose)
//   P1: per row c: NTT_N1 over r in LDS (12 stages max, 128 KB tile),
//       epilogue multiplies element k1 by w^(k1*c)   (tw_full table)
//   T1: C[k1][c] = C1[c][k1]
//   P2: per row k1: NTT_N2 over c in LDS (+ 1/n scale for iNTT)
//   T2: out[k2*N1+k1] = D[k1][k2]
// Identical integer results to the radix-2 path (same DFT), parity-pinned
// against the oracle.

// tiled fe4 transpose, 32x32 tiles (+1 pad column kills LDS bank conflicts)
__global__ void __launch_bounds__(256)
k_transpose_fe4(const fe4 *__restrict__ src, fe4 *__restrict__ dst,
                uint32_t R, uint32_t C) {
    __shared__ fe4 tile[32][33];
    uint32_t c0 = blockIdx.x * 32, r0 = blockIdx.y * 32;
    uint32_t tx = threadIdx.x & 31, ty = threadIdx.x >> 5;  // 8 rows/pass
    for (uint32_t dy = ty; dy < 32; dy += 8)
        tile[dy][tx] = src[(size_t)(r0 + dy) * C + c0 + tx];
    __syncthreads();
    for (uint32_t dy = ty; dy < 32; dy += 8)
        dst[(size_t)(c0 + dy) * R + r0 + tx] = tile[tx][dy];
}

// one row NTT of length M = 2^logM fully in LDS.
// tw_row: M/2 twiddles of the size-M transform.
// tw_full: if non-null, epilogue multiplies element k by tw_full[k*blockIdx.x]
//          (the four-step inter-NTT twiddle; index < n always).
// scale: if non-null, epilogue multiplies by *scale (iNTT 1/n, Montgomery).
__global__ void __launch_bounds__(1024)
k_ntt_row(fe4 *__restrict__ data, int logM, const fe4 *__restrict__ tw_row,
          const fe4 *__restrict__ tw_full, const fe4 *__restrict__ scale) {
    // static 128 KiB LDS (max row length 4096 fe4); gfx950 has 160 KiB/CU —
    // one block/CU, 8 waves, streaming kernel
    __shared__ fe4 smem[4096];
    const uint32_t M = 1u << logM;
    fe4 *row = data + (size_t)blockIdx.x * M;
    for (uint32_t i = threadIdx.x; i < M; i += blockDim.x) {
        uint32_t j = __brev(i) >> (32 - logM);
        smem[j] = row[i];
    }
    __syncthreads();
    for (int s = 1; s <= logM; s++) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t t = threadIdx.x; t < (M >> 1); t += blockDim.x) {
            uint32_t j = t & (half - 1);
            uint32_t idx = ((t >> (s - 1)) << s) + j;
            fe4 u = smem[idx];
            fe4 v = mont_mul<Fr>(smem[idx + half], tw_row[j << (logM - s)]);
            smem[idx] = mod_add<Fr>(u, v);
            smem[idx + half] = mod_sub<Fr>(u, v);
        }
        __syncthreads();
    }
    uint64_t c = blockIdx.x;
    for (uint32_t k = threadIdx.x; k < M; k += blockDim.x) {
        fe4 x = smem[k];
        if (tw_full) x = mont_mul<Fr>(x, tw_full[(size_t)k * c]);
        if (scale) x = mont_mul<Fr>(x, *scale);
        row[k] = x;
    }
}

// ---- scale by n^-1 (iNTT) ----
__global__ void k_ntt_scale(fe4 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe4 ninv{{bn254::FR_INV_POW2_MONT[logn][0], bn254::FR_INV_POW2_MONT[logn][1],
              bn254::FR_INV_POW2_MONT[logn][2], bn254::FR_INV_POW2_MONT[logn][3]}};
    a[i] = mont_mul<Fr>(a[i], ninv);
}

}  // namespace em
