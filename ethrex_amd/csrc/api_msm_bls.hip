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
// api_msm_bls.hip — BLS12-381 G1 MSM ABI (blob-KZG / EIP-2537 G1).
#include "msm_api_impl.h"

extern "C" int ethrex_mi355_bls_msm_plan_create(size_t n, em_bls_msm_plan **plan) {
    return msm_create_t(n, (msm_plan_t<BlsG1> **)plan);
}
extern "C" int ethrex_mi355_bls_msm_plan_destroy(em_bls_msm_plan *p) {
    return msm_destroy_t((msm_plan_t<BlsG1> *)p);
}
extern "C" int ethrex_mi355_bls_msm_upload_points(em_bls_msm_plan *p,
                                                  const uint8_t *points96) {
    return msm_upload_points_t((msm_plan_t<BlsG1> *)p, points96);
}
extern "C" int ethrex_mi355_bls_msm_gen_points(em_bls_msm_plan *p,
                                               uint64_t start) {
    return msm_gen_points_t((msm_plan_t<BlsG1> *)p, start);
}
extern "C" int ethrex_mi355_bls_msm_download_points(em_bls_msm_plan *p,
                                                    uint8_t *out96) {
    return msm_download_points_t((msm_plan_t<BlsG1> *)p, out96);
}
extern "C" int ethrex_mi355_bls_msm_upload_scalars(em_bls_msm_plan *p,
                                                   const uint8_t *scalars32) {
    return msm_upload_scalars_t((msm_plan_t<BlsG1> *)p, scalars32);
}
extern "C" int ethrex_mi355_bls_msm_run(em_bls_msm_plan *p, uint8_t out[96]) {
    return msm_run_inner_t((msm_plan_t<BlsG1> *)p, out, 0);
}
extern "C" int ethrex_mi355_bls_msm_run_async(em_bls_msm_plan *p,
                                              uint8_t out[96]) {
    return msm_run_async_t((msm_plan_t<BlsG1> *)p, out);
}
extern "C" int ethrex_mi355_bls_msm_sync(em_bls_msm_plan *p) {
    return msm_sync_t((msm_plan_t<BlsG1> *)p);
}
extern "C" int ethrex_mi355_bls_msm_run_partial(em_bls_msm_plan *p,
                                                uint8_t out[144]) {
    return msm_run_inner_t((msm_plan_t<BlsG1> *)p, out, 1);
}
extern "C" int ethrex_mi355_bls_msm_precompute(em_bls_msm_plan *p) {
    return msm_precompute_t((msm_plan_t<BlsG1> *)p);
}
extern "C" int ethrex_mi355_bls_msm_last_times(em_bls_msm_plan *p,
                                               double times_ms[5]) {
    if (!p || !times_ms) return EM_ERR_INPUT;
    memcpy(times_ms, p->last_ms, sizeof p->last_ms);
    return EM_OK;
}
extern "C" int ethrex_mi355_bls12381_g1_msm(const uint8_t *points96,
                                            const uint8_t *scalars32, size_t n,
                                            uint8_t out[96]) {
    if (!points96 || !scalars32 || !out || n == 0) return EM_ERR_INPUT;
    em_bls_msm_plan *p = nullptr;
    int rc = ethrex_mi355_bls_msm_plan_create(n, &p);
    if (rc) return rc;
    rc = ethrex_mi355_bls_msm_upload_points(p, points96);
    if (!rc) rc = ethrex_mi355_bls_msm_upload_scalars(p, scalars32);
    if (!rc) rc = ethrex_mi355_bls_msm_run(p, out);
    ethrex_mi355_bls_msm_plan_destroy(p);
    return rc;
}
extern "C" int ethrex_mi355_bls12381_g1_add(const uint8_t p1[96],
                                            const uint8_t p2[96],
                                            uint8_t out[96]) {
    if (!p1 || !p2 || !out) return EM_ERR_INPUT;
    return run_single(k_bls_g1_add_single, p1, 96, p2, 96, out, 96);
}

extern "C" int ethrex_mi355_bls12381_g1_mul(const uint8_t point[96],
                                            const uint8_t scalar[32],
                                            uint8_t out[96]) {
    if (!point || !scalar || !out) return EM_ERR_INPUT;
    return run_single(k_bls_g1_mul_single, point, 96, scalar, 32, out, 96);
}
extern "C" int ethrex_mi355_bls12381_g1_combine(const uint8_t *jacobians144,
                                                size_t count, uint8_t out[96]) {
    if (!jacobians144 || !out || count == 0) return EM_ERR_INPUT;
    int rc = require_gpu();
    if (rc) return rc;
    uint8_t *d_in, *d_out;
    HIP_TRY(hipMalloc(&d_in, 144 * count));
    HIP_TRY(hipMalloc(&d_out, 96));
    HIP_TRY(hipMemcpy(d_in, jacobians144, 144 * count, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_g1_combine<BlsG1>, dim3(1), dim3(64), 0, 0, d_in,
                       count, d_out);
    HIP_TRY(hipMemcpy(out, d_out, 96, hipMemcpyDeviceToHost));
    (void)hipFree(d_in);
    (void)hipFree(d_out);
    return EM_OK;
}
