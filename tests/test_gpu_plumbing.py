"""GPU end-to-end plumbing: the Mi355Backend driving the gfx950 MSM/NTT
core through the full pull-loop protocol against the mock coordinator —
the drop-in path a Rust ProverBackend impl would exercise,
with the proof's MSM result parity-checked against the oracle."""
import hashlib
import json
import os

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

        # recompute the expected proof with the oracle: prove() runs the
        # wrap-shaped composition MSM(points, NTT(seeded vector)) with the
        # NTT output handed to the MSM on device (sp1.rs:122-134 flow)
        input_data = {"batch": 0, "blocks": [1, 2, 3]}
        seed = int.from_bytes(hashlib.sha256(json.dumps(
            input_data, sort_keys=True).encode()).digest()[:8], "little")
        m = 1 << 14
        pts = oracle_mod.gen_points(0, m)
        rc, fwd = oracle_mod.fr_ntt(oracle_mod.gen_fr(seed, m), m, False)
        assert rc == 0
        rc, want_msm = oracle_mod.g1_msm(pts, fwd, m)
        assert rc == 0 and proof_bytes[:64] == want_msm
        assert proof_bytes[64:] == hashlib.sha256(fwd).digest()
    finally:
        coord.stop()


def test_mi355_backend_statement_from_real_witness(oracle_mod):
    """the proof binds to a STATEMENT derived from
    real witness bytes — the hoodi fixture witness is validated on the
    GPU (batched-keccak node hashing + trie linking) and the MSM input
    derives from the statement commitment, parity-checked via the
    oracle."""
    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    from ethrex_amd import witness as W
    from ethrex_amd.prover import ExecBackend, Mi355Backend

    fx = os.path.join(os.path.dirname(__file__), "golden",
                      "witness_hoodi_1265656.json.gz")
    state, headers, fbn = W.load_witness_fixture(fx)
    input_data = {
        "batch": 7,
        "witness": {"state": ["0x" + s.hex() for s in state],
                    "headers": ["0x" + h.hex() for h in headers]},
        "first_block_number": fbn,
    }
    backend = Mi355Backend(msm_log2=14, ntt_log2=12)
    proof = backend.prove(input_data, None)
    st = proof["statement"]
    # GPU-hashed witness statement == CPU-hashed statement (ExecBackend,
    # the reference's CPU exec path) — keccak engines agree bit-exactly
    st_cpu = ExecBackend().execute(input_data)
    assert st == st_cpu
    assert st["initial_state_root"] == (
        "4bec425c34f89aeb56c78586d586c76044bcc428e7c34ab7c72d389c39ff3eaf")
    # the MSM input really derives from the statement commitment through
    # the wrap composition: MSM(points, NTT(statement-seeded vector))
    seed = int.from_bytes(bytes.fromhex(st["commitment"])[:8], "little")
    m = 1 << 14
    pts = oracle_mod.gen_points(0, m)
    rc, fwd = oracle_mod.fr_ntt(oracle_mod.gen_fr(seed, m), m, False)
    assert rc == 0
    rc, want_msm = oracle_mod.g1_msm(pts, fwd, m)
    assert rc == 0 and proof["msm"] == want_msm
    out = backend.to_proof_bytes(proof, None)
    pb = bytes(out["Proof"]["proof"])
    assert len(pb) == 128 and pb[96:] == bytes.fromhex(st["commitment"])


def test_gpu_trie_root_level_synchronized(oracle_mod):
    """§8f row 4 second half: MPT root with every tree level hashed as
    one batched GPU keccak launch — equals the CPU-hashed root and the
    recursive reference on random account-shaped pairs."""
    import random
    import sys

    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    from ethrex_amd import trie, witness
    from ethrex_amd.prover import Mi355Backend
    sys.path.insert(0, os.path.dirname(__file__))
    from mpt_reference import mpt_root

    be = Mi355Backend()
    rng = random.Random(11)
    pairs = {}
    for i in range(2000):
        k = bytes(rng.randrange(256) for _ in range(32))
        pairs[k] = trie.account_leaf(i, i * 7, witness.EMPTY_TRIE_HASH,
                                     bytes(32))
    got = trie.trie_root(pairs, be._gpu_hash_batch)
    assert got == trie.trie_root(pairs, witness.cpu_hash_batch)
    assert got == mpt_root(pairs)

