// ============================================================================
// Batched Keccak-256 on gfx950 (the witness/statement
// hashing step feeding the prover — ExecBackend::execute_core's keccak/trie
// hashing; the reference computes these one at a time through hand-written
// x86/ARM asm, crates/common/crypto/keccak/mod.rs + keccak1600-x86_64.s).
//
// One LANE per message: the 25-lane u64 state lives in VGPRs (~50 VGPRs),
// keccak-f1600 is ~1900 64-bit VALU ops per permutation, and a 4096-lane
// launch already fills a CU's issue; messages are independent so the batch
// shape is embarrassingly parallel.  Original Keccak padding (0x01 ... 0x80),
// rate 136 — Ethereum keccak256, NOT SHA3-256.
//
// Parity: ethrex_amd/keccak.py (pure-Python restatement pinned by the
// canonical vectors keccak256("")=c5d246..., keccak256("abc")=4e0365...);
// GPU tests compare bit-exactly across rate-boundary lengths.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>

namespace em {

__device__ __constant__ static const uint64_t KRC[24] = {
    0x0000000000000001ull, 0x0000000000008082ull, 0x800000000000808Aull,
    0x8000000080008000ull, 0x000000000000808Bull, 0x0000000080000001ull,
    0x8000000080008081ull, 0x8000000000008009ull, 0x000000000000008Aull,
    0x0000000000000088ull, 0x0000000080008009ull, 0x000000008000000Aull,
    0x000000008000808Bull, 0x800000000000008Bull, 0x8000000000008089ull,
    0x8000000000008003ull, 0x8000000000008002ull, 0x8000000000000080ull,
    0x000000000000800Aull, 0x800000008000000Aull, 0x8000000080008081ull,
    0x8000000000008080ull, 0x0000000080000001ull, 0x8000000080008008ull};

// rotation offsets r[x][y] (column-major x + 5y, as keccak.py's _ROT)
__device__ __constant__ static const int KROT[25] = {
    0,  36, 3,  41, 18,   // x = 0
    1,  44, 10, 45, 2,    // x = 1
    62, 6,  43, 15, 61,   // x = 2
    28, 55, 25, 21, 56,   // x = 3
    27, 20, 39, 8,  14};  // x = 4

__device__ __forceinline__ uint64_t krotl(uint64_t v, int n) {
    return n == 0 ? v : (v << n) | (v >> (64 - n));
}

__device__ void keccak_f1600(uint64_t a[25]) {
    for (int r = 0; r < 24; r++) {
        uint64_t c[5], d[5], b[25];
#pragma unroll
        for (int x = 0; x < 5; x++)
            c[x] = a[x] ^ a[x + 5] ^ a[x + 10] ^ a[x + 15] ^ a[x + 20];
#pragma unroll
        for (int x = 0; x < 5; x++)
            d[x] = c[(x + 4) % 5] ^ krotl(c[(x + 1) % 5], 1);
#pragma unroll
        for (int x = 0; x < 5; x++)
#pragma unroll
            for (int y = 0; y < 5; y++) {
                b[y + 5 * ((2 * x + 3 * y) % 5)] =
                    krotl(a[x + 5 * y] ^ d[x], KROT[5 * x + y]);
            }
#pragma unroll
        for (int x = 0; x < 5; x++)
#pragma unroll
            for (int y = 0; y < 5; y++)
                a[x + 5 * y] =
                    b[x + 5 * y] ^ (~b[(x + 1) % 5 + 5 * y] &
                                    b[(x + 2) % 5 + 5 * y]);
        a[0] ^= KRC[r];
    }
}

__device__ __forceinline__ uint64_t load_le64(const uint8_t *p) {
    if (((uintptr_t)p & 7) == 0) return *(const uint64_t *)p;
    uint64_t v = 0;
#pragma unroll
    for (int k = 0; k < 8; k++) v |= (uint64_t)p[k] << (8 * k);
    return v;
}

// one message per lane; offsets[n] delimits message i = [offs[i], offs[i+1])
__global__ void __launch_bounds__(256)
k_keccak256_batch(const uint8_t *__restrict__ msgs,
                  const uint64_t *__restrict__ offs, size_t n,
                  uint8_t *__restrict__ out) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t lo = offs[i], hi = offs[i + 1];
    const uint8_t *p = msgs + lo;
    uint64_t rem = hi - lo;
    uint64_t st[25];
#pragma unroll
    for (int k = 0; k < 25; k++) st[k] = 0;
    while (rem >= 136) {
#pragma unroll
        for (int j = 0; j < 17; j++) st[j] ^= load_le64(p + 8 * j);
        keccak_f1600(st);
        p += 136;
        rem -= 136;
    }
    // final block with the original-Keccak pad10*1 (0x01 ... 0x80)
    uint64_t blk[17];
#pragma unroll
    for (int j = 0; j < 17; j++) blk[j] = 0;
    uint8_t *bb = (uint8_t *)blk;
    for (uint64_t k = 0; k < rem; k++) bb[k] = p[k];
    bb[rem] ^= 0x01;
    bb[135] ^= 0x80;
#pragma unroll
    for (int j = 0; j < 17; j++) st[j] ^= blk[j];
    keccak_f1600(st);
    uint64_t *o = (uint64_t *)(out + 32 * i);
#pragma unroll
    for (int j = 0; j < 4; j++) o[j] = st[j];
}

}  // namespace em
