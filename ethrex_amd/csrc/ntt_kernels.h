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

// ---- scale by n^-1 (iNTT) ----
__global__ void k_ntt_scale(fe4 *__restrict__ a, size_t n, int logn) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fe4 ninv{{bn254::FR_INV_POW2_MONT[logn][0], bn254::FR_INV_POW2_MONT[logn][1],
              bn254::FR_INV_POW2_MONT[logn][2], bn254::FR_INV_POW2_MONT[logn][3]}};
    a[i] = mont_mul<Fr>(a[i], ninv);
}

}  // namespace em
