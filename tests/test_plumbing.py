"""CPU plumbing parity: the ProofData protocol, assignment semantics and
pull loop against the mock coordinator (BASELINE.md plumbing config —
"ethrex's existing CPU prover path, plumbing, no GPU" — using the
reference's own mock-prover shape, ExecBackend)."""
import threading
import time

import pytest

from ethrex_amd.coordinator import MockCoordinator
from ethrex_amd.prover import BackendError, ExecBackend, ProverClient


@pytest.fixture
def coord():
    c = MockCoordinator(commit_hash="deadbeef").start()
    yield c
    c.stop()


def test_pull_prove_submit(coord):
    for i in range(3):
        coord.add_batch(i, {"batch": i, "blocks": [i * 10]})
    client = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)])
    n = 0
    for _ in range(10):
        n += client.poll_once()
        if n >= 3:
            break
    assert sorted(client.proved) == [0, 1, 2]
    assert set(coord.proofs) == {(0, "Exec"), (1, "Exec"), (2, "Exec")}
    # sentinel proof bytes shape (exec.rs:53-60)
    assert coord.proofs[(0, "Exec")]["Proof"]["proof"] == [0]


def test_version_mismatch(coord):
    coord.add_batch(0, {"batch": 0})
    client = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)],
                          commit_hash="wrong")
    with pytest.raises(BackendError, match="version"):
        client.poll_once()


def test_no_work_empty_response(coord):
    client = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)])
    assert client.poll_once() == 0


def test_assignment_timeout_reassigns():
    c = MockCoordinator(timeout_s=0.2).start()
    try:
        c.add_batch(7, {"batch": 7})

        class StallingBackend(ExecBackend):
            def prove(self, input_data, fmt):
                raise OSError("prover died mid-proof")

        dead = ProverClient(StallingBackend(), [("127.0.0.1", c.port)])
        assert dead.poll_once() == 0  # took assignment, never submitted
        live = ProverClient(ExecBackend(), [("127.0.0.1", c.port)])
        assert live.poll_once() == 0  # still assigned to the dead prover
        time.sleep(0.25)
        assert live.poll_once() == 1  # reassigned after timeout
    finally:
        c.stop()


def test_duplicate_submit_noop(coord):
    coord.add_batch(1, {"batch": 1})
    a = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)])
    assert a.poll_once() == 1
    stored = coord.proofs[(1, "Exec")]
    # second client re-submits directly (restart-safety: no-op)
    from ethrex_amd.prover import _round_trip, proof_output
    ack = _round_trip("127.0.0.1", coord.port, {"ProofSubmit": {
        "id": 1, "proof": proof_output("Exec", b"\x99")}})
    assert "ProofSubmitACK" in ack
    assert coord.proofs[(1, "Exec")] == stored  # first proof kept


def test_two_provers_split_batches(coord):
    """Batch-level data parallelism: N independent pull clients, one
    coordinator (the reference's only parallelism over this path)."""
    for i in range(6):
        coord.add_batch(i, {"batch": i})
    a = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)])
    b = ProverClient(ExecBackend(), [("127.0.0.1", coord.port)])
    ta = threading.Thread(target=lambda: [a.poll_once() for _ in range(8)])
    tb = threading.Thread(target=lambda: [b.poll_once() for _ in range(8)])
    ta.start(); tb.start(); ta.join(); tb.join()
    assert len(coord.proofs) == 6
    assert sorted(a.proved + b.proved) == list(range(6))
