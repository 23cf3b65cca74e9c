"""GPU end-to-end plumbing: the Mi355Backend driving the gfx950 MSM/NTT
core through the full pull-loop protocol against the mock coordinator —
the drop-in path a Rust ProverBackend impl would exercise,
with the proof's MSM result parity-checked against the oracle."""
import hashlib
import json

import pytest

pytestmark = pytest.mark.gpu


def test_mi355_backend_proves_through_coordinator(oracle_mod):
    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    from ethrex_amd.coordinator import MockCoordinator
    from ethrex_amd.prover import Mi355Backend, ProverClient

    coord = MockCoordinator().start()
    try:
        coord.add_batch(0, {"batch": 0, "blocks": [1, 2, 3]})
        backend = Mi355Backend(msm_log2=14, ntt_log2=12)
        client = ProverClient(backend, [("127.0.0.1", coord.port)])
        n = 0
        for _ in range(5):
            n += client.poll_once()
            if n:
                break
        assert n == 1 and client.proved == [0]
        stored = coord.proofs[(0, "Exec")]
        proof_bytes = bytes(stored["Proof"]["proof"])
        assert len(proof_bytes) == 96  # 64 B MSM affine + 32 B NTT digest

        # recompute the expected proof with the oracle (same derivation)
        input_data = {"batch": 0, "blocks": [1, 2, 3]}
        seed = int.from_bytes(hashlib.sha256(json.dumps(
            input_data, sort_keys=True).encode()).digest()[:8], "little")
        m = 1 << 14
        pts = oracle_mod.gen_points(0, m)
        scs = oracle_mod.gen_fr(seed, m)
        rc, want_msm = oracle_mod.g1_msm(pts, scs, m)
        assert rc == 0 and proof_bytes[:64] == want_msm
        k = 1 << 12
        rc, want_ntt = oracle_mod.fr_ntt(oracle_mod.gen_fr(seed + 1, k), k, False)
        assert rc == 0
        assert proof_bytes[64:] == hashlib.sha256(want_ntt).digest()
    finally:
        coord.stop()
