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
// em_api_common.h — shared internal plumbing for the ABI translation units
// (split for parallel compilation: the template-heavy MSM machinery costs
// ~10 min of clang per curve at -O3; one TU per curve builds with make -j).
#pragma once
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <type_traits>

// thread-local last-error buffer (defined in api_core.hip)
extern thread_local std::string g_last_err;

static inline int hip_fail(hipError_t e, const char *where) {
    g_last_err = std::string(where) + ": " + hipGetErrorString(e);
    return 3;  // EM_ERR_HIP
}

#define HIP_TRY(call)                                    \
    do {                                                 \
        hipError_t _e = (call);                          \
        if (_e != hipSuccess) return hip_fail(_e, #call); \
    } while (0)

static inline int require_gpu() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess || n == 0) {
        g_last_err = "no HIP device visible (MI355X required; no CPU fallback)";
        return 3;  // EM_ERR_HIP
    }
    return 0;
}

static inline uint32_t blocks_for(size_t n, int bs) {
    return (uint32_t)((n + bs - 1) / bs);
}

