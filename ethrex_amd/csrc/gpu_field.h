// ============================================================================
// Device 4x64-limb Montgomery field arithmetic for BN254 Fq / Fr — gfx950.
//
// Written for CDNA4: big-integer modular mul is VALU work (no MFMA — it is
// not a dense fp contraction).  64x64->128 products lower to v_mad_u64_u32
// chains on amdgcn; everything is fully unrolled so limbs live in VGPRs and
// the compiler can schedule the carry chains.
//
// Semantics restated from the reference's ark-bn254 path
// (crates/common/crypto/provider.rs:247-318): 4x64 little-endian limbs,
// Montgomery R = 2^256, inputs to/from big-endian canonical bytes.
// Implemented independently from oracle/bn254_oracle.c (CIOS here, SOS
// there) so the parity comparison does not share bugs.
// ============================================================================
#pragma once
#include <hip/hip_runtime.h>
#include "bn254_constants_dev.h"

namespace em {

using u32 = uint32_t;
using u64 = uint64_t;
using u128 = unsigned __int128;

struct fe4 {
    u64 v[4];
};

using Fq = bn254::Fq;
using Fr = bn254::Fr;

__device__ __host__ __forceinline__ bool fe_is_zero(const fe4 &a) {
    return (a.v[0] | a.v[1] | a.v[2] | a.v[3]) == 0;
}

// a >= b ?
__device__ __host__ __forceinline__ bool fe_geq(const fe4 &a, const fe4 &b) {
#pragma unroll
    for (int i = 3; i >= 0; i--) {
        if (a.v[i] != b.v[i]) return a.v[i] > b.v[i];
    }
    return true;
}

// ---- CIOS Montgomery multiplication, 32-bit-limb form ----
// On CDNA4 the 64-bit-limb form compiles to long VCC carry chains
// (v_addc + mandatory s_nop wait states — measured 863 instructions); the
// 32-bit-limb form below maps every step onto carry-free u64 arithmetic
// (v_mad_u64_u32 + v_lshl_add_u64, zero s_nop / zero v_addc — 569
// instructions), because each `c += (u64)a32*b32 + t32` fits u64 exactly.
// Inputs < MOD  =>  output < MOD (standard CIOS bound; the to_mont case
// with a < 2^256 and b = R2 < MOD also stays within the guard limb).

template <typename F>
__device__ __host__ __forceinline__ fe4 mont_mul(const fe4 &A, const fe4 &B) {
    u32 a[8], b[8], n[8];
#pragma unroll
    for (int i = 0; i < 4; i++) {
        a[2 * i] = (u32)A.v[i];
        a[2 * i + 1] = (u32)(A.v[i] >> 32);
        b[2 * i] = (u32)B.v[i];
        b[2 * i + 1] = (u32)(B.v[i] >> 32);
        n[2 * i] = (u32)F::MOD[i];
        n[2 * i + 1] = (u32)(F::MOD[i] >> 32);
    }
    const u32 n0inv32 = (u32)F::N0INV;  // -MOD^-1 mod 2^32
    u32 t[10] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int i = 0; i < 8; i++) {
        u64 c = 0;
#pragma unroll
        for (int j = 0; j < 8; j++) {
            c += (u64)a[i] * b[j] + t[j];
            t[j] = (u32)c;
            c >>= 32;
        }
        c += t[8];
        t[8] = (u32)c;
        t[9] = (u32)(c >> 32);
        u32 m = t[0] * n0inv32;
        c = (u64)m * n[0] + t[0];
        c >>= 32;
#pragma unroll
        for (int j = 1; j < 8; j++) {
            c += (u64)m * n[j] + t[j];
            t[j - 1] = (u32)c;
            c >>= 32;
        }
        c += t[8];
        t[7] = (u32)c;
        t[8] = t[9] + (u32)(c >> 32);
        t[9] = 0;
    }
    fe4 out{{(u64)t[0] | ((u64)t[1] << 32), (u64)t[2] | ((u64)t[3] << 32),
             (u64)t[4] | ((u64)t[5] << 32), (u64)t[6] | ((u64)t[7] << 32)}};
    // final conditional subtraction (t[8] is 0 or 1)
    bool ge = t[8] != 0;
    if (!ge) ge = fe_geq(out, fe4{{F::MOD[0], F::MOD[1], F::MOD[2], F::MOD[3]}});
    if (ge) {
        u128 bor = 0;
#pragma unroll
        for (int i = 0; i < 4; i++) {
            u128 t2 = (u128)out.v[i] - F::MOD[i] - bor;
            out.v[i] = (u64)t2;
            bor = (t2 >> 64) & 1;
        }
    }
    return out;
}

// to Montgomery form; input may be ANY 256-bit value — CIOS with b = R2 < MOD
// and a < 2^256 keeps the accumulator within the guard limbs and fully
// reduces, mirroring ark's from_be_bytes_mod_order for 32-byte inputs.
template <typename F>
__device__ __host__ __forceinline__ fe4 to_mont(const fe4 &a) {
    return mont_mul<F>(a, fe4{{F::R2[0], F::R2[1], F::R2[2], F::R2[3]}});
}

template <typename F>
__device__ __host__ __forceinline__ fe4 from_mont(const fe4 &a) {
    return mont_mul<F>(a, fe4{{1, 0, 0, 0}});
}

// ---- big-endian byte conversion ----

__device__ __host__ __forceinline__ u64 bswap64(u64 x) {
#ifdef __HIP_DEVICE_COMPILE__
    return __builtin_bswap64(x);
#else
    return __builtin_bswap64(x);
#endif
}

// load 32 big-endian bytes -> 4x64 LE limbs (no reduction)
__device__ __host__ __forceinline__ fe4 fe_from_be(const uint8_t *b) {
    const u64 *w = (const u64 *)b;  // byte buffers are 8-aligned by ABI contract
    fe4 r;
    r.v[3] = bswap64(w[0]);
    r.v[2] = bswap64(w[1]);
    r.v[1] = bswap64(w[2]);
    r.v[0] = bswap64(w[3]);
    return r;
}

__device__ __host__ __forceinline__ void fe_to_be(uint8_t *b, const fe4 &x) {
    u64 *w = (u64 *)b;
    w[0] = bswap64(x.v[3]);
    w[1] = bswap64(x.v[2]);
    w[2] = bswap64(x.v[1]);
    w[3] = bswap64(x.v[0]);
}

}  // namespace em
