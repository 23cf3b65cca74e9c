"""CPU tests for the proof-object wire integration:
keccak (pinned canonical vectors), the verifyBatches selector, the
consecutive-batch collection semantics of l1_proof_sender.rs:298-318, and
the Solidity ABI encoding of l1_proof_sender.rs:514-556.
"""
from ethrex_amd.keccak import keccak256, selector
from ethrex_amd.proof_sender import (ProofSender, collect_ready_batches,
                                     decode_verify_batches_calldata,
                                     encode_verify_batches_calldata,
                                     VERIFY_BATCHES_SIG)
from ethrex_amd.prover import proof_output


def test_keccak256_canonical_vectors():
    assert keccak256(b"").hex() == ("c5d2460186f7233c927e7db2dcc703c0"
                                    "e500b653ca82273b7bfad8045d85a470")
    assert keccak256(b"abc").hex() == ("4e03657aea45a94fc7d47ba826c8d667"
                                       "c0d1e6e33a64a036ec44f58fa12d6c45")
    # rate-boundary cases (136-byte rate)
    assert len(keccak256(b"x" * 136)) == 32
    assert keccak256(b"x" * 135) != keccak256(b"x" * 136)


def test_selector_shape():
    s = selector(VERIFY_BATCHES_SIG)
    assert len(s) == 4
    # the well-known transfer selector pins the selector derivation
    assert selector("transfer(address,uint256)").hex() == "a9059cbb"


def test_collect_consecutive_stops_at_gap():
    store = {}
    for b in (1, 2, 4):
        store[(b, "SP1")] = proof_output("SP1", bytes([b]))
        store[(b, "RISC0")] = proof_output("RISC0", bytes([b, b]))
    ready = collect_ready_batches(store, 0, 5, ("SP1", "RISC0"))
    assert [b for b, _ in ready] == [1, 2]  # 3 missing -> stop
    # a batch missing ONE needed type is not ready
    store[(3, "SP1")] = proof_output("SP1", b"\x03")
    ready = collect_ready_batches(store, 0, 5, ("SP1", "RISC0"))
    assert [b for b, _ in ready] == [1, 2]
    # bounded by last_committed
    ready = collect_ready_batches(store, 0, 1, ("SP1", "RISC0"))
    assert [b for b, _ in ready] == [1]


def test_calldata_roundtrip_and_layout():
    batches = [
        (7, {"SP1": proof_output("SP1", b"\xaa" * 33),
             "RISC0": proof_output("RISC0", b"\xbb" * 7)}),
        (8, {"SP1": proof_output("SP1", b"\xcc" * 64)}),
    ]
    data = encode_verify_batches_calldata(7, batches)
    assert data[:4] == selector(VERIFY_BATCHES_SIG)
    assert len(data) % 32 == 4  # selector + 32-byte words
    first, (risc0, sp1, tdx) = decode_verify_batches_calldata(data)
    assert first == 7
    # array order per l1_proof_sender.rs:539-545: risc0, sp1, tdx
    assert risc0 == [b"\xbb" * 7, b""]      # batch 8 has no RISC0 proof
    assert sp1 == [b"\xaa" * 33, b"\xcc" * 64]
    assert tdx == [b"", b""]


def test_proof_sender_flow():
    sent = []
    ps = ProofSender(("SP1",), on_chain_proposer="0xproposer",
                     send=lambda target, calldata: sent.append(
                         (target, calldata)))
    ps.last_committed = 3
    assert ps.verify_and_send() == 0  # nothing stored
    ps.store_proof(1, "SP1", proof_output("SP1", b"p1"))
    ps.store_proof(2, "SP1", proof_output("SP1", b"p2"))
    assert ps.verify_and_send() == 2
    assert ps.last_verified == 2
    target, data = sent[0]
    assert target == "0xproposer"
    first, (_r, sp1, _t) = decode_verify_batches_calldata(data)
    assert first == 1 and sp1 == [b"p1", b"p2"]
    # verification cursor advanced: resend only new batches
    ps.store_proof(3, "SP1", proof_output("SP1", b"p3"))
    assert ps.verify_and_send() == 1
    first, (_r, sp1, _t) = decode_verify_batches_calldata(sent[1][1])
    assert first == 3 and sp1 == [b"p3"]


def test_proof_with_public_values_variant():
    """ProverOutput::ProofWithPublicValues carries the same ProofBytes
    (prover.rs:64-100); the sender uses proof_bytes().proof."""
    out = {"ProofWithPublicValues": {
        "proof_bytes": {"prover_type": "SP1", "proof": list(b"zz")},
        "public_values": list(b"pv")}}
    data = encode_verify_batches_calldata(1, [(1, {"SP1": out})])
    _, (_r, sp1, _t) = decode_verify_batches_calldata(data)
    assert sp1 == [b"zz"]


def test_calldata_matches_hand_derived_golden():
    """Byte-for-byte check against an INDEPENDENTLY hand-derived calldata
    vector (the round-trip test could not catch an offset-
    layout misconception shared by encoder and decoder).  The expected
    bytes below are constructed word-by-word from the Solidity ABI spec
    (head/tail encoding, element offsets relative to the start of each
    dynamic array's data area), mirroring send_verify_batches_tx
    (l1_proof_sender.rs:514-556): two batches, SP1 proofs only.
    """
    p1, p2 = b"\xaa\xbb\xcc", b"\x11" * 33
    batches = [
        (5, {"SP1": {"Proof": {"proof": list(p1)}}}),
        (6, {"SP1": {"Proof": {"proof": list(p2)}}}),
    ]
    got = encode_verify_batches_calldata(5, batches)

    def w(v):  # one 32-byte big-endian word
        return v.to_bytes(32, "big")

    # keccak("verifyBatches(uint256,bytes[],bytes[],bytes[])")[:4] — the
    # keccak itself is pinned by canonical vectors in test_keccak256_*
    want = bytes.fromhex("9711f750")
    want += w(5)            # uint256 firstBatchNumber
    # three bytes[] head offsets, relative to start of args (head = 4 words)
    want += w(128)          # risc0Proofs: right after the head
    want += w(128 + 160)    # sp1Proofs: after risc0 tail (160 B, below)
    want += w(288 + 256)    # tdxProofs: after sp1 tail (256 B, below)
    # risc0Proofs = ["", ""]: count, two element offsets relative to the
    # word after the count (2 offset words = 64 B), two empty bytes
    # elements (a zero length word each) => 5 words = 160 B
    want += w(2) + w(64) + w(96) + w(0) + w(0)
    # sp1Proofs = [p1 (3 B -> 2 words), p2 (33 B -> 3 words)]:
    # count, offsets 64 and 64+64, then len+data+pad for each => 256 B
    want += w(2) + w(64) + w(128)
    want += w(3) + p1 + b"\x00" * 29
    want += w(33) + p2 + b"\x00" * 31
    # tdxProofs = ["", ""]
    want += w(2) + w(64) + w(96) + w(0) + w(0)
    assert got == want
