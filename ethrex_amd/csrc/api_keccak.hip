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
// api_keccak.hip — batched keccak-256 (witness/trie hashing).
#include "em_api_common.h"
#include "../../include/ethrex_mi355.h"
#include "keccak_kernels.h"

using namespace em;

// ======================= batched keccak-256 =======================
// witness/statement hashing (the reference's
// crates/common/crypto/keccak asm path, called per-node in trie hashing).

struct em_keccak_plan {
    size_t max_bytes = 0, max_n = 0, n = 0;
    uint8_t *d_msgs = nullptr;
    uint64_t *d_offs = nullptr;
    uint8_t *d_out = nullptr;
    hipEvent_t ev[2];
    double last_ms = 0;
};

extern "C" int ethrex_mi355_keccak_plan_create(size_t max_bytes, size_t max_n,
                                               em_keccak_plan **plan) {
    if (!plan || max_n == 0) return EM_ERR_INPUT;
    int rc = require_gpu();
    if (rc) return rc;
    auto *p = new em_keccak_plan();
    p->max_bytes = max_bytes;
    p->max_n = max_n;
    hipError_t e = hipMalloc((void **)&p->d_msgs, max_bytes ? max_bytes : 1);
    if (e == hipSuccess) e = hipMalloc((void **)&p->d_offs, (max_n + 1) * 8);
    if (e == hipSuccess) e = hipMalloc((void **)&p->d_out, max_n * 32);
    for (int i = 0; i < 2 && e == hipSuccess; i++)
        e = hipEventCreate(&p->ev[i]);
    if (e != hipSuccess) {
        (void)hipFree(p->d_msgs);
        (void)hipFree(p->d_offs);
        (void)hipFree(p->d_out);
        delete p;
        return hip_fail(e, "keccak_plan_create");
    }
    *plan = p;
    return EM_OK;
}

extern "C" int ethrex_mi355_keccak_plan_destroy(em_keccak_plan *p) {
    if (!p) return EM_ERR_INPUT;
    (void)hipFree(p->d_msgs);
    (void)hipFree(p->d_offs);
    (void)hipFree(p->d_out);
    delete p;
    return EM_OK;
}

extern "C" int ethrex_mi355_keccak_upload(em_keccak_plan *p,
                                          const uint8_t *msgs,
                                          const uint64_t *offsets, size_t n) {
    if (!p || !offsets || n == 0 || n > p->max_n) return EM_ERR_INPUT;
    if (offsets[n] > p->max_bytes) return EM_ERR_INPUT;
    if (offsets[n] > 0 && !msgs) return EM_ERR_INPUT;
    if (offsets[n] > 0)
        HIP_TRY(hipMemcpy(p->d_msgs, msgs, offsets[n],
                          hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(p->d_offs, offsets, (n + 1) * 8,
                      hipMemcpyHostToDevice));
    p->n = n;
    return EM_OK;
}

extern "C" int ethrex_mi355_keccak_run(em_keccak_plan *p) {
    if (!p || p->n == 0) return EM_ERR_INPUT;
    HIP_TRY(hipEventRecord(p->ev[0], 0));
    hipLaunchKernelGGL(k_keccak256_batch, dim3(blocks_for(p->n, 256)),
                       dim3(256), 0, 0, p->d_msgs, p->d_offs, p->n, p->d_out);
    HIP_TRY(hipEventRecord(p->ev[1], 0));
    HIP_TRY(hipDeviceSynchronize());
    float ms;
    HIP_TRY(hipEventElapsedTime(&ms, p->ev[0], p->ev[1]));
    p->last_ms = ms;
    return EM_OK;
}

extern "C" int ethrex_mi355_keccak_download(em_keccak_plan *p, uint8_t *out) {
    if (!p || !out || p->n == 0) return EM_ERR_INPUT;
    HIP_TRY(hipMemcpy(out, p->d_out, p->n * 32, hipMemcpyDeviceToHost));
    return EM_OK;
}

extern "C" int ethrex_mi355_keccak_last_ms(em_keccak_plan *p, double *ms) {
    if (!p || !ms) return EM_ERR_INPUT;
    *ms = p->last_ms;
    return EM_OK;
}

/* one-shot convenience (PCIe-inclusive) */
extern "C" int ethrex_mi355_keccak256_batch(const uint8_t *msgs,
                                            const uint64_t *offsets, size_t n,
                                            uint8_t *out32) {
    if (!offsets || !out32 || n == 0) return EM_ERR_INPUT;
    em_keccak_plan *p = nullptr;
    int rc = ethrex_mi355_keccak_plan_create(offsets[n], n, &p);
    if (rc) return rc;
    rc = ethrex_mi355_keccak_upload(p, msgs, offsets, n);
    if (!rc) rc = ethrex_mi355_keccak_run(p);
    if (!rc) rc = ethrex_mi355_keccak_download(p, out32);
    ethrex_mi355_keccak_plan_destroy(p);
    return rc;
}
