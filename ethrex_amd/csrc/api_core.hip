// ============================================================================
// ethrex_mi355 C-ABI implementation — MI355X-native BN254 MSM/NTT core.
//
// See include/ethrex_mi355.h for the boundary contract (mirrors the in-repo
// ZisK accelerator FFI convention, crates/guest-program/src/crypto/zisk.rs:71-137)
// and DESIGN.md for the kernel design.  Threading: the backend is called
// from ONE actor on a blocking thread (crates/prover/src/prover.rs:241-251),
// so plans are not internally locked.
//
// NO CPU FALLBACK: every compute entry point requires a visible GPU and
// returns EM_ERR_HIP otherwise.
// ============================================================================
// api_core.hip — version/device plumbing + deterministic input generation.
#include "em_api_common.h"
#include "../../include/ethrex_mi355.h"
#include "gpu_field.h"

using namespace em;

thread_local std::string g_last_err;

extern "C" const char *ethrex_mi355_version(void) { return "0.1.0-gfx950"; }

extern "C" const char *ethrex_mi355_last_error(void) { return g_last_err.c_str(); }

extern "C" int ethrex_mi355_device_count(int *count) {
    if (!count) return EM_ERR_INPUT;
    hipError_t e = hipGetDeviceCount(count);
    if (e != hipSuccess) {
        *count = 0;
        return hip_fail(e, "hipGetDeviceCount");
    }
    return EM_OK;
}

extern "C" int ethrex_mi355_set_device(int device) {
    HIP_TRY(hipSetDevice(device));
    return EM_OK;
}


// ============================ input generation ============================
// Product-side restatement of BASELINE.md's deterministic input scheme
// (splitmix64-seeded xoshiro256++, rejection to [0, r), 254-bit mask).
// Parity-tested against the oracle's independent restatement.

namespace {

struct Xosh {
    uint64_t s[4];
};

uint64_t splitmix64_next(uint64_t &x) {
    uint64_t z = (x += 0x9e3779b97f4a7c15ull);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    return z ^ (z >> 31);
}

uint64_t rotl64(uint64_t x, int k) { return (x << k) | (x >> (64 - k)); }

uint64_t xosh_next(Xosh &g) {
    uint64_t r = rotl64(g.s[0] + g.s[3], 23) + g.s[0];
    uint64_t t = g.s[1] << 17;
    g.s[2] ^= g.s[0];
    g.s[3] ^= g.s[1];
    g.s[1] ^= g.s[2];
    g.s[0] ^= g.s[3];
    g.s[2] ^= t;
    g.s[3] = rotl64(g.s[3], 45);
    return r;
}

}  // namespace

extern "C" void ethrex_mi355_gen_fr(uint64_t seed, size_t n, uint8_t *out) {
    Xosh g;
    uint64_t sm = seed;
    for (int i = 0; i < 4; i++) g.s[i] = splitmix64_next(sm);
    const fe4 rmod{{Fr::MOD[0], Fr::MOD[1], Fr::MOD[2], Fr::MOD[3]}};
    for (size_t i = 0; i < n; i++) {
        fe4 s;
        do {
            s.v[0] = xosh_next(g);
            s.v[1] = xosh_next(g);
            s.v[2] = xosh_next(g);
            s.v[3] = xosh_next(g) & 0x3fffffffffffffffull;
        } while (fe_geq(s, rmod));
        fe_to_be(out + 32 * i, s);
    }
}

// same scheme for BLS12-381 Fr (255-bit mask, reject >= r_bls)
extern "C" void ethrex_mi355_bls_gen_fr(uint64_t seed, size_t n, uint8_t *out) {
    Xosh g;
    uint64_t sm = seed;
    for (int i = 0; i < 4; i++) g.s[i] = splitmix64_next(sm);
    const fe4 rmod{{bn254::FrB::MOD[0], bn254::FrB::MOD[1], bn254::FrB::MOD[2],
                    bn254::FrB::MOD[3]}};
    for (size_t i = 0; i < n; i++) {
        fe4 s;
        do {
            s.v[0] = xosh_next(g);
            s.v[1] = xosh_next(g);
            s.v[2] = xosh_next(g);
            s.v[3] = xosh_next(g) & 0x7fffffffffffffffull;
        } while (fe_geq(s, rmod));
        fe_to_be(out + 32 * i, s);
    }
}
