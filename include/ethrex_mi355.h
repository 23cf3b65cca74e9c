/* ============================================================================
 * ethrex_mi355 — C-ABI boundary of the MI355X-native BN254 MSM/NTT prover
 * core.
 *
 * This is the drop-in seam a Rust host binds behind ethrex's `ProverBackend`
 * trait (reference: crates/prover/src/backend/mod.rs:87-153; the backend's
 * `prove` drives these entry points).  The FFI convention mirrors the
 * in-repo ZisK accelerator ABI (crates/guest-program/src/crypto/zisk.rs:71-137):
 *   - `int` status returns (0 = OK, nonzero = error),
 *   - fixed-size byte structs, 8-byte aligned,
 *   - caller-allocated out-parameters.
 * The Rust-side binding a maintainer would add is shown in INTEGRATION.md.
 *
 * Byte encodings (reference semantics, crates/common/crypto/provider.rs:247-318):
 *   - G1 affine point: 64 bytes big-endian x||y; (0,0) encodes the identity
 *     on input and output (EIP-197; crates/vm/levm/src/precompiles.rs:792-795).
 *     Coordinates are parsed `from_be_bytes_mod_order` (reduced mod p) and
 *     validated on-curve (off-curve => EM_ERR_POINT).
 *   - Scalar / Fr element: 32 bytes big-endian; scalars are reduced mod r
 *     exactly as ark's `from_be_bytes_mod_order` does.  NTT elements must be
 *     canonical (< r) => EM_ERR_INPUT otherwise.
 *   - G1 Jacobian partial (multi-GPU exchange payload): 96 bytes big-endian
 *     X||Y||Z canonical form; Z = 0 encodes the identity.
 *
 * Every compute entry point REQUIRES the MI355X GPU path: there is no CPU
 * fallback anywhere behind this ABI (EM_ERR_HIP if no device).
 * ==========================================================================*/
#ifndef ETHREX_MI355_H
#define ETHREX_MI355_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
    EM_OK = 0,
    EM_ERR_POINT = 1,   /* G1 point not on curve (provider.rs InvalidPoint) */
    EM_ERR_INPUT = 2,   /* bad size / non-canonical Fr element / null ptr   */
    EM_ERR_HIP = 3,     /* HIP runtime failure or no gfx950 device          */
};

/* library identification / device selection */
const char *ethrex_mi355_version(void);
int ethrex_mi355_device_count(int *count);
int ethrex_mi355_set_device(int device);
/* last HIP error string (for diagnostics after EM_ERR_HIP) */
const char *ethrex_mi355_last_error(void);

/* ---- single-op entry points (zisk.rs:100-102 zkvm_bn254_g1_add/mul
 *      mirror; run on the GPU; used for on-device semantics parity) ---- */
int ethrex_mi355_bn254_g1_add(const uint8_t p1[64], const uint8_t p2[64],
                              uint8_t out[64]);
int ethrex_mi355_bn254_g1_mul(const uint8_t point[64], const uint8_t scalar[32],
                              uint8_t out[64]);

/* ---- one-shot hot-path entry points (host buffers in, host buffer out;
 *      the drop-in calls a Rust `ProverBackend::prove` makes) ---- */
int ethrex_mi355_bn254_g1_msm(const uint8_t *points64, const uint8_t *scalars32,
                              size_t n, uint8_t out[64]);
int ethrex_mi355_bn254_fr_ntt(uint8_t *elems32, size_t n, int inverse);

/* ---- planned API: device-resident buffers for repeated runs (bench /
 *      pipeline use; inputs stay in HBM between runs) ---- */
typedef struct em_msm_plan em_msm_plan;
typedef struct em_ntt_plan em_ntt_plan;

int ethrex_mi355_msm_plan_create(size_t n, em_msm_plan **plan);
int ethrex_mi355_msm_plan_destroy(em_msm_plan *plan);
/* upload host-side affine points (64 B BE each); validates on device */
int ethrex_mi355_msm_upload_points(em_msm_plan *plan, const uint8_t *points64);
/* generate P_i = (start+i+1)*G directly in HBM (deterministic inputs) */
int ethrex_mi355_msm_gen_points(em_msm_plan *plan, uint64_t start);
/* download the plan's points back as 64 B BE affine (parity checks) */
int ethrex_mi355_msm_download_points(em_msm_plan *plan, uint8_t *out64);
int ethrex_mi355_msm_upload_scalars(em_msm_plan *plan, const uint8_t *scalars32);
/* full MSM -> affine result */
int ethrex_mi355_msm_run(em_msm_plan *plan, uint8_t out[64]);
/* shard partial -> Jacobian 96 B (multi-GPU: AllGather these, then combine) */
int ethrex_mi355_msm_run_partial(em_msm_plan *plan, uint8_t out[96]);
/* pipelined MSM step: enqueues the step and returns; the sort chain of the
 * NEXT run_async overlaps this step's bucket/reduction chain on a second
 * HIP stream (the proving loop runs many MSMs back-to-back).  `out` is
 * filled by the time ethrex_mi355_msm_sync — or the depth-2 backpressure of
 * a later run_async — returns; keep it valid until then. */
int ethrex_mi355_msm_run_async(em_msm_plan *plan, uint8_t out[64]);
/* drain all pipelined steps and deliver their results */
int ethrex_mi355_msm_sync(em_msm_plan *plan);
/* pipelined shard step: run_async delivering the 96-B Jacobian partial
 * (the multi-GPU exchange payload) instead of the affine result */
int ethrex_mi355_msm_run_partial_async(em_msm_plan *plan, uint8_t out[96]);
/* deliver the OLDEST pending pipelined step only (blocks on its compute
 * chain; later enqueued steps keep running).  The N>1 exchange loop calls
 * this to AllGather+combine step k while the GPU computes step k+1. */
int ethrex_mi355_msm_wait_one(em_msm_plan *plan);
/* combine Jacobian partials on the GPU -> affine result */
int ethrex_mi355_bn254_g1_combine(const uint8_t *jacobians96, size_t count,
                                  uint8_t out[64]);
/* combine the N>1 exchange payload (world-size 96-B Jacobian partials) on
 * the HOST — boundary glue; avoids touching GPU streams mid-pipeline */
int ethrex_mi355_bn254_g1_combine_cpu(const uint8_t *jacobians96,
                                      size_t count, uint8_t out[64]);
/* wrap-pipeline handoff (sp1.rs:122-134 flow): take this plan's n scalars
 * from an NTT plan's device-resident output (ethrex_mi355_ntt_device_data)
 * at element `offset` — the composed MSM+NTT step stays on-device */
int ethrex_mi355_msm_scalars_from_ntt(em_msm_plan *plan, const void *ntt_data,
                                      uint64_t offset);
/* per-phase HIP-event timings of the last run, milliseconds:
 * [0]=digits+sort, [1]=bucket accumulation, [2]=bucket reduction,
 * [3]=window combine + affine, [4]=total */
int ethrex_mi355_msm_last_times(em_msm_plan *plan, double times_ms[5]);
/* combine partials reusing the plan's device buffers (per-step exchange) */
int ethrex_mi355_msm_combine(em_msm_plan *plan, const uint8_t *jacobians96,
                             size_t count, uint8_t out[64]);

/* ---- native MPT structure builder (host; §8f row 4 witness-generation
 * speedup).  Fixed 32-byte sorted distinct keys (the hashed-key state /
 * storage trie shape); per-level >= 32-B node encodings come out in the
 * batched-keccak layout, are hashed on the GPU by the caller, and the
 * hashes patch back before the next (shallower) level encodes.  Levels
 * run from mpt_max_depth down to 0; the root is always hashed. ---- */
typedef struct em_mpt em_mpt;
int ethrex_mi355_mpt_create(const uint8_t *keys32, const uint8_t *vals,
                            const uint64_t *val_offs, size_t n, em_mpt **out);
int ethrex_mi355_mpt_destroy(em_mpt *t);
int ethrex_mi355_mpt_max_depth(em_mpt *t, int *depth);
int ethrex_mi355_mpt_level_encode(em_mpt *t, int depth, uint8_t *buf,
                                  uint64_t *offs, size_t buf_cap,
                                  size_t max_n, size_t *n_hash);
int ethrex_mi355_mpt_level_set_hashes(em_mpt *t, int depth,
                                      const uint8_t *h32, size_t n_hash);
int ethrex_mi355_mpt_root(em_mpt *t, uint8_t out[32]);

/* ---- batched Keccak-256 (witness/statement hashing; original Keccak
 * padding, Ethereum keccak256).  offsets[n+1] delimits message i as
 * [offsets[i], offsets[i+1]); out32 receives 32 bytes per message. ---- */
typedef struct em_keccak_plan em_keccak_plan;
int ethrex_mi355_keccak256_batch(const uint8_t *msgs, const uint64_t *offsets,
                                 size_t n, uint8_t *out32);
int ethrex_mi355_keccak_plan_create(size_t max_bytes, size_t max_n,
                                    em_keccak_plan **plan);
int ethrex_mi355_keccak_plan_destroy(em_keccak_plan *plan);
int ethrex_mi355_keccak_upload(em_keccak_plan *plan, const uint8_t *msgs,
                               const uint64_t *offsets, size_t n);
int ethrex_mi355_keccak_run(em_keccak_plan *plan);
int ethrex_mi355_keccak_download(em_keccak_plan *plan, uint8_t *out32);
int ethrex_mi355_keccak_last_ms(em_keccak_plan *plan, double *ms);

int ethrex_mi355_ntt_plan_create(size_t n, em_ntt_plan **plan);
int ethrex_mi355_ntt_plan_destroy(em_ntt_plan *plan);
int ethrex_mi355_ntt_upload(em_ntt_plan *plan, const uint8_t *elems32);
int ethrex_mi355_ntt_run(em_ntt_plan *plan, int inverse);
int ethrex_mi355_ntt_download(em_ntt_plan *plan, uint8_t *elems32);
/* [0]=bit-reverse, [1]=butterfly stages total, [2]=total */
int ethrex_mi355_ntt_last_times(em_ntt_plan *plan, double times_ms[3]);
/* device pointer to the current transform output (packed 4x64 Montgomery)
 * for the on-device MSM handoff (ethrex_mi355_msm_scalars_from_ntt) */
int ethrex_mi355_ntt_device_data(em_ntt_plan *plan, const void **ptr,
                                 size_t *n);

/* ---- deterministic input generation (host-side; the product restatement
 *      of BASELINE.md's xoshiro256++/splitmix64 scheme; parity-tested
 *      against the oracle's independent restatement) ---- */
void ethrex_mi355_gen_fr(uint64_t seed, size_t n, uint8_t *out32);

/* ==== BLS12-381 G1 (the blob-KZG commitment MSM
 *      of crates/common/crypto/kzg.rs:208-230 and the EIP-2537 G1 MSM of
 *      provider.rs:620-634 / bls_blst.rs).  Semantics = the in-tree blst
 *      path: 48-byte big-endian CANONICAL coordinates (non-canonical =>
 *      EM_ERR_INPUT), (0,0) identity, on-curve check, r-subgroup check on
 *      MSM inputs (=> EM_ERR_POINT), scalars = full 256-bit integers (no
 *      reduction).  Affine point = 96 B x||y; Jacobian partial = 144 B
 *      X||Y||Z canonical, Z=0 identity. ==== */

int ethrex_mi355_bls12381_g1_add(const uint8_t p1[96], const uint8_t p2[96],
                                 uint8_t out[96]);
int ethrex_mi355_bls12381_g1_mul(const uint8_t point[96],
                                 const uint8_t scalar[32], uint8_t out[96]);
int ethrex_mi355_bls12381_g1_msm(const uint8_t *points96,
                                 const uint8_t *scalars32, size_t n,
                                 uint8_t out[96]);
int ethrex_mi355_bls12381_g1_combine(const uint8_t *jacobians144, size_t count,
                                     uint8_t out[96]);

typedef struct em_bls_msm_plan em_bls_msm_plan;
int ethrex_mi355_bls_msm_plan_create(size_t n, em_bls_msm_plan **plan);
int ethrex_mi355_bls_msm_plan_destroy(em_bls_msm_plan *plan);
int ethrex_mi355_bls_msm_upload_points(em_bls_msm_plan *plan,
                                       const uint8_t *points96);
int ethrex_mi355_bls_msm_gen_points(em_bls_msm_plan *plan, uint64_t start);
int ethrex_mi355_bls_msm_download_points(em_bls_msm_plan *plan, uint8_t *out96);
int ethrex_mi355_bls_msm_upload_scalars(em_bls_msm_plan *plan,
                                        const uint8_t *scalars32);
/* build the fixed-base table (2^(12w)*P_i, c-kzg KZG_PRECOMPUTE-style):
 * the setup points are fixed across blobs, so subsequent runs collapse the
 * window dimension.  Requires n <= 65536; invalidated by new points. */
int ethrex_mi355_bls_msm_precompute(em_bls_msm_plan *plan);
int ethrex_mi355_bls_msm_run(em_bls_msm_plan *plan, uint8_t out[96]);
int ethrex_mi355_bls_msm_run_partial(em_bls_msm_plan *plan, uint8_t out[144]);
int ethrex_mi355_bls_msm_run_async(em_bls_msm_plan *plan, uint8_t out[96]);
int ethrex_mi355_bls_msm_sync(em_bls_msm_plan *plan);

/* ---- BLS12-381 G2 (EIP-2537 / blst semantics, bls_blst.rs:338-441):
 * 192-byte points x.c0||x.c1||y.c0||y.c1 (canonical 48-B BE coords over
 * Fp2 = Fp[u]/(u^2+1)), (0,0,0,0) = identity; add checks on-curve only,
 * MSM additionally enforces the r-subgroup check per point.  Jacobian
 * exchange payload: 288 B (X||Y||Z, each c0||c1). ---- */
typedef struct em_bls_g2_msm_plan em_bls_g2_msm_plan;
int ethrex_mi355_bls12381_g2_add(const uint8_t p1[192], const uint8_t p2[192],
                                 uint8_t out[192]);
int ethrex_mi355_bls12381_g2_mul(const uint8_t point[192],
                                 const uint8_t scalar[32], uint8_t out[192]);
int ethrex_mi355_bls12381_g2_msm(const uint8_t *points192,
                                 const uint8_t *scalars32, size_t n,
                                 uint8_t out[192]);
int ethrex_mi355_bls_g2_msm_plan_create(size_t n, em_bls_g2_msm_plan **plan);
int ethrex_mi355_bls_g2_msm_plan_destroy(em_bls_g2_msm_plan *plan);
int ethrex_mi355_bls_g2_msm_upload_points(em_bls_g2_msm_plan *plan,
                                          const uint8_t *points192);
int ethrex_mi355_bls_g2_msm_gen_points(em_bls_g2_msm_plan *plan,
                                       uint64_t start);
int ethrex_mi355_bls_g2_msm_download_points(em_bls_g2_msm_plan *plan,
                                            uint8_t *out192);
int ethrex_mi355_bls_g2_msm_upload_scalars(em_bls_g2_msm_plan *plan,
                                           const uint8_t *scalars32);
int ethrex_mi355_bls_g2_msm_run(em_bls_g2_msm_plan *plan, uint8_t out[192]);
int ethrex_mi355_bls_g2_msm_run_async(em_bls_g2_msm_plan *plan,
                                      uint8_t out[192]);
int ethrex_mi355_bls_g2_msm_sync(em_bls_g2_msm_plan *plan);
int ethrex_mi355_bls_g2_msm_run_partial(em_bls_g2_msm_plan *plan,
                                        uint8_t out[288]);
int ethrex_mi355_bls_g2_msm_last_times(em_bls_g2_msm_plan *plan,
                                       double times_ms[5]);
int ethrex_mi355_bls_msm_last_times(em_bls_msm_plan *plan, double times_ms[5]);

/* n elements uniform in [0, r_bls) (canonical blob field elements) */
void ethrex_mi355_bls_gen_fr(uint64_t seed, size_t n, uint8_t *out32);

#ifdef __cplusplus
}
#endif

#endif /* ETHREX_MI355_H */
