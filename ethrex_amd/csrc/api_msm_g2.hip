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
// api_msm_g2.hip — BLS12-381 G2 MSM ABI (EIP-2537 G2 ops over Fp2).
#include "msm_api_impl.h"

// ---- BLS12-381 G2 plan + one-shot ABI (EIP-2537 192-byte points) ----
extern "C" int ethrex_mi355_bls_g2_msm_plan_create(size_t n,
                                                   em_bls_g2_msm_plan **plan) {
    return msm_create_t(n, (msm_plan_t<BlsG2> **)plan);
}
extern "C" int ethrex_mi355_bls_g2_msm_plan_destroy(em_bls_g2_msm_plan *p) {
    return msm_destroy_t((msm_plan_t<BlsG2> *)p);
}
extern "C" int ethrex_mi355_bls_g2_msm_upload_points(em_bls_g2_msm_plan *p,
                                                     const uint8_t *pts192) {
    return msm_upload_points_t((msm_plan_t<BlsG2> *)p, pts192);
}
extern "C" int ethrex_mi355_bls_g2_msm_gen_points(em_bls_g2_msm_plan *p,
                                                  uint64_t start) {
    return msm_gen_points_t((msm_plan_t<BlsG2> *)p, start);
}
extern "C" int ethrex_mi355_bls_g2_msm_download_points(em_bls_g2_msm_plan *p,
                                                       uint8_t *out192) {
    return msm_download_points_t((msm_plan_t<BlsG2> *)p, out192);
}
extern "C" int ethrex_mi355_bls_g2_msm_upload_scalars(em_bls_g2_msm_plan *p,
                                                      const uint8_t *s32) {
    return msm_upload_scalars_t((msm_plan_t<BlsG2> *)p, s32);
}
extern "C" int ethrex_mi355_bls_g2_msm_run(em_bls_g2_msm_plan *p,
                                           uint8_t out[192]) {
    return msm_run_inner_t((msm_plan_t<BlsG2> *)p, out, 0);
}
extern "C" int ethrex_mi355_bls_g2_msm_run_async(em_bls_g2_msm_plan *p,
                                                 uint8_t out[192]) {
    return msm_run_async_t((msm_plan_t<BlsG2> *)p, out);
}
extern "C" int ethrex_mi355_bls_g2_msm_sync(em_bls_g2_msm_plan *p) {
    return msm_sync_t((msm_plan_t<BlsG2> *)p);
}
extern "C" int ethrex_mi355_bls_g2_msm_run_partial(em_bls_g2_msm_plan *p,
                                                   uint8_t out[288]) {
    return msm_run_inner_t((msm_plan_t<BlsG2> *)p, out, 1);
}
extern "C" int ethrex_mi355_bls_g2_msm_last_times(em_bls_g2_msm_plan *p,
                                                  double times_ms[5]) {
    if (!p) return EM_ERR_INPUT;
    for (int i = 0; i < 5; i++) times_ms[i] = ((msm_plan_t<BlsG2> *)p)->last_ms[i];
    return EM_OK;
}
extern "C" int ethrex_mi355_bls12381_g2_msm(const uint8_t *points192,
                                            const uint8_t *scalars32, size_t n,
                                            uint8_t out[192]) {
    if (!points192 || !scalars32 || !out || n == 0) return EM_ERR_INPUT;
    em_bls_g2_msm_plan *p = nullptr;
    int rc = ethrex_mi355_bls_g2_msm_plan_create(n, &p);
    if (rc) return rc;
    rc = ethrex_mi355_bls_g2_msm_upload_points(p, points192);
    if (!rc) rc = ethrex_mi355_bls_g2_msm_upload_scalars(p, scalars32);
    if (!rc) rc = ethrex_mi355_bls_g2_msm_run(p, out);
    ethrex_mi355_bls_g2_msm_plan_destroy(p);
    return rc;
}
extern "C" int ethrex_mi355_bls12381_g2_add(const uint8_t p1[192],
                                            const uint8_t p2[192],
                                            uint8_t out[192]) {
    return run_single(k_bls_g2_add_single, p1, 192, p2, 192, out, 192);
}

extern "C" int ethrex_mi355_bls12381_g2_mul(const uint8_t point[192],
                                            const uint8_t scalar[32],
                                            uint8_t out[192]) {
    return run_single(k_bls_g2_mul_single, point, 192, scalar, 32, out, 192);
}
