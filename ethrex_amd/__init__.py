"""ethrex_amd — MI355X-native BN254 MSM/NTT prover core (host side).

Python host layer over the C-ABI library `libethrex_mi355.so`
(include/ethrex_mi355.h).  The reference's host for this path is compiled
Rust behind the `ProverBackend` trait (crates/prover/src/backend/mod.rs:87-153);
the Rust binding a maintainer would add over the same C-ABI is shown in
INTEGRATION.md.  This package mirrors that interface for the plumbing tests
and the bench harness.

The product path FAILS LOUDLY if the HIP library is missing — there is no
CPU fallback anywhere (DESIGN.md "Oracle discipline").
"""
from .lib import (EM_OK, EM_ERR_POINT, EM_ERR_INPUT, EM_ERR_HIP,
                  MsmPlan, NttPlan, BlsMsmPlan, device_count, set_device,
                  version, g1_add, g1_mul, g1_msm, fr_ntt, g1_combine,
                  g1_combine_cpu, gen_fr,
                  bls_g1_add, bls_g1_mul, bls_g1_msm, bls_g1_combine,
                  bls_gen_fr, bls_g2_add, bls_g2_mul, bls_g2_msm,
                  BlsG2MsmPlan, keccak256_batch, KeccakPlan, last_error)

__all__ = [
    "EM_OK", "EM_ERR_POINT", "EM_ERR_INPUT", "EM_ERR_HIP",
    "MsmPlan", "NttPlan", "BlsMsmPlan", "device_count", "set_device",
    "version", "g1_add", "g1_mul", "g1_msm", "fr_ntt", "g1_combine",
    "g1_combine_cpu", "gen_fr",
    "bls_g1_add", "bls_g1_mul", "bls_g1_msm", "bls_g1_combine", "bls_gen_fr",
    "bls_g2_add", "bls_g2_mul", "bls_g2_msm", "BlsG2MsmPlan",
    "keccak256_batch", "KeccakPlan",
    "last_error",
]
