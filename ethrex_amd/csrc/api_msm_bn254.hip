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
// api_msm_bn254.hip — BN254 G1 MSM ABI (plan + one-shot + single ops).
#include "msm_api_impl.h"

extern "C" int ethrex_mi355_msm_plan_create(size_t n, em_msm_plan **plan) {
    return msm_create_t(n, (msm_plan_t<Bn254G1> **)plan);
}
extern "C" int ethrex_mi355_msm_plan_destroy(em_msm_plan *p) {
    return msm_destroy_t((msm_plan_t<Bn254G1> *)p);
}
extern "C" int ethrex_mi355_msm_upload_points(em_msm_plan *p,
                                              const uint8_t *points64) {
    return msm_upload_points_t((msm_plan_t<Bn254G1> *)p, points64);
}
extern "C" int ethrex_mi355_msm_gen_points(em_msm_plan *p, uint64_t start) {
    return msm_gen_points_t((msm_plan_t<Bn254G1> *)p, start);
}
extern "C" int ethrex_mi355_msm_download_points(em_msm_plan *p, uint8_t *out64) {
    return msm_download_points_t((msm_plan_t<Bn254G1> *)p, out64);
}
extern "C" int ethrex_mi355_msm_upload_scalars(em_msm_plan *p,
                                               const uint8_t *scalars32) {
    return msm_upload_scalars_t((msm_plan_t<Bn254G1> *)p, scalars32);
}
extern "C" int ethrex_mi355_msm_run(em_msm_plan *p, uint8_t out[64]) {
    return msm_run_inner_t((msm_plan_t<Bn254G1> *)p, out, 0);
}
extern "C" int ethrex_mi355_msm_run_async(em_msm_plan *p, uint8_t out[64]) {
    return msm_run_async_t((msm_plan_t<Bn254G1> *)p, out);
}
extern "C" int ethrex_mi355_msm_sync(em_msm_plan *p) {
    return msm_sync_t((msm_plan_t<Bn254G1> *)p);
}
/* deliver the OLDEST pending pipelined step without draining the pipeline
 * (N>1 exchange overlap: AllGather step k while the GPU computes k+1) */
extern "C" int ethrex_mi355_msm_wait_one(em_msm_plan *p) {
    return msm_wait_one_t((msm_plan_t<Bn254G1> *)p);
}
/* pipelined shard step: like run_async but delivers the 96-B Jacobian
 * partial (multi-GPU: the NEXT step's sort chain overlaps this step's
 * compute even across the AllGather + combine exchange). */
extern "C" int ethrex_mi355_msm_run_partial_async(em_msm_plan *p,
                                                  uint8_t out[96]) {
    return msm_run_async_t((msm_plan_t<Bn254G1> *)p, out, 1);
}
extern "C" int ethrex_mi355_msm_run_partial(em_msm_plan *p, uint8_t out[96]) {
    return msm_run_inner_t((msm_plan_t<Bn254G1> *)p, out, 1);
}
extern "C" int ethrex_mi355_msm_last_times(em_msm_plan *p, double times_ms[5]) {
    if (!p || !times_ms) return EM_ERR_INPUT;
    memcpy(times_ms, p->last_ms, sizeof p->last_ms);
    return EM_OK;
}

// combine Jacobian partials reusing the plan's buffers (no per-call
// hipMalloc: the N>1 exchange runs this every step)
extern "C" int ethrex_mi355_msm_combine(em_msm_plan *p,
                                        const uint8_t *jacobians96,
                                        size_t count, uint8_t out[64]) {
    if (!p || !jacobians96 || !out || count == 0 || count * 96 > p->n * 64)
        return EM_ERR_INPUT;
    HIP_TRY(hipMemcpy(p->d_scratch, jacobians96, 96 * count,
                      hipMemcpyHostToDevice));
    hipLaunchKernelGGL((k_g1_combine<Bn254G1>), dim3(1), dim3(64), 0, 0,
                       p->d_scratch, count, p->d_out);
    HIP_TRY(hipMemcpy(out, p->d_out, 64, hipMemcpyDeviceToHost));
    return EM_OK;
}
// ============================ one-shot MSMs ============================

extern "C" int ethrex_mi355_bn254_g1_msm(const uint8_t *points64,
                                         const uint8_t *scalars32, size_t n,
                                         uint8_t out[64]) {
    if (!points64 || !scalars32 || !out || n == 0) return EM_ERR_INPUT;
    em_msm_plan *p = nullptr;
    int rc = ethrex_mi355_msm_plan_create(n, &p);
    if (rc) return rc;
    rc = ethrex_mi355_msm_upload_points(p, points64);
    if (!rc) rc = ethrex_mi355_msm_upload_scalars(p, scalars32);
    if (!rc) rc = ethrex_mi355_msm_run(p, out);
    ethrex_mi355_msm_plan_destroy(p);
    return rc;
}
extern "C" int ethrex_mi355_bn254_g1_add(const uint8_t p1[64],
                                         const uint8_t p2[64], uint8_t out[64]) {
    if (!p1 || !p2 || !out) return EM_ERR_INPUT;
    return run_single(k_g1_add_single, p1, 64, p2, 64, out, 64);
}

extern "C" int ethrex_mi355_bn254_g1_mul(const uint8_t point[64],
                                         const uint8_t scalar[32],
                                         uint8_t out[64]) {
    if (!point || !scalar || !out) return EM_ERR_INPUT;
    return run_single(k_g1_mul_single, point, 64, scalar, 32, out, 64);
}
extern "C" int ethrex_mi355_bn254_g1_combine(const uint8_t *jacobians96,
                                             size_t count, uint8_t out[64]) {
    if (!jacobians96 || !out || count == 0) return EM_ERR_INPUT;
    int rc = require_gpu();
    if (rc) return rc;
    uint8_t *d_in, *d_out;
    HIP_TRY(hipMalloc(&d_in, 96 * count));
    HIP_TRY(hipMalloc(&d_out, 64));
    HIP_TRY(hipMemcpy(d_in, jacobians96, 96 * count, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_g1_combine<Bn254G1>, dim3(1), dim3(64), 0, 0, d_in,
                       count, d_out);
    HIP_TRY(hipMemcpy(out, d_out, 64, hipMemcpyDeviceToHost));
    (void)hipFree(d_in);
    (void)hipFree(d_out);
    return EM_OK;
}

// HOST-side combine of the N>1 exchange payload: `count` = world size (a
// handful of 96-B Jacobian partials, one per rank).  This is boundary glue
// like the delivery-time Horner — the MSM compute stays on the GPU; doing
// these few adds on the host avoids interposing a kernel + 2 copies on the
// default stream while pipelined steps are in flight.
extern "C" int ethrex_mi355_bn254_g1_combine_cpu(const uint8_t *jacobians96,
                                                 size_t count,
                                                 uint8_t out[64]) {
    if (!jacobians96 || !out || count == 0) return EM_ERR_INPUT;
    g1jT<Bn254G1> acc = g1_inf9<Bn254G1>();
    for (size_t i = 0; i < count; i++) {
        g1jT<Bn254G1> t;
        if (!g1_jac_from_be9<Bn254G1>(t, jacobians96 + 96 * i)) continue;
        acc = g1_add9<Bn254G1>(acc, t);
    }
    g1_to_affine_be9<Bn254G1>(out, acc);
    return EM_OK;
}

// wrap-pipeline handoff: take this plan's n scalars from an NTT plan's
// device-resident output (packed fe4m Montgomery, via
// ethrex_mi355_ntt_device_data) starting at element `offset` — the
// composed 2^26 MSM+NTT "wrap" step stays on-device end to end.
namespace {
__global__ void k_scalars_from_fe4m(const fe4 *__restrict__ src,
                                    fe4 *__restrict__ dst, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    // fe4m -> fe9 (pure repacking) -> canonical (< r) -> packed u64x4,
    // exactly the value layout k_parse_scalars produces
    fe9 m = fe9_from_u64x4(src[i].v);
    fe9 c = from_mont9<Fr9T>(m);
    fe4 out;
    fe9_to_u64x4(out.v, c);
    dst[i] = out;
}
}  // namespace

extern "C" int ethrex_mi355_msm_scalars_from_ntt(em_msm_plan *plan,
                                                 const void *ntt_data,
                                                 uint64_t offset) {
    auto *p = (msm_plan_t<Bn254G1> *)plan;
    if (!p || !ntt_data) return EM_ERR_INPUT;
    int rc0 = msm_sync_t(p);
    if (rc0) return rc0;
    hipLaunchKernelGGL(k_scalars_from_fe4m, dim3(blocks_for(p->n, 256)),
                       dim3(256), 0, 0, (const fe4 *)ntt_data + offset,
                       p->d_scalars, p->n);
    HIP_TRY(hipDeviceSynchronize());
    p->have_scalars = true;
    return EM_OK;
}
