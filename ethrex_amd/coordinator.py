"""Mock proof coordinator speaking the reference's ProofData protocol.

Wire format mirrors ethrex's serde-JSON-over-TCP protocol
(crates/common/types/prover.rs:119-159; server loop
crates/l2/sequencer/proof_coordinator.rs:101-199):
  - externally-tagged enum JSON: {"InputRequest": {...}}, "VersionMismatch",
    {"InputResponse": {...}}, {"ProofSubmit": {...}}, {"ProofSubmitACK": {...}}
  - one JSON message per connection in each direction (connect, send, recv,
    close — prover.rs:289-305 connect_to_prover_server_wr semantics)
  - assignment map keyed (batch_id, prover_type) with timeout reassignment
    (proof_coordinator.rs:52-57,149-199)
  - duplicate proof submission is a no-op (docs/l2 distributed_proving.md)

Test infrastructure for plumbing parity (BASELINE.md plumbing config):
the real consumer of proofs is out of scope.
"""
import json
import socket
import threading
import time


class MockCoordinator:
    def __init__(self, commit_hash="deadbeef", timeout_s=600.0):
        self.commit_hash = commit_hash
        self.timeout_s = timeout_s
        self.batches = {}          # id -> input dict
        self.format = "Groth16"    # ProofFormat default (prover.rs:103-110)
        self.assigned = {}         # (id, prover_type) -> assign time
        self.proofs = {}           # (id, prover_type) -> ProverOutput dict
        self._lock = threading.Lock()
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind(("127.0.0.1", 0))
        self._srv.listen(16)
        self.port = self._srv.getsockname()[1]
        self._stop = False
        self._thread = threading.Thread(target=self._serve, daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._stop = True
        try:
            socket.create_connection(("127.0.0.1", self.port), timeout=1).close()
        except OSError:
            pass
        self._thread.join(timeout=5)
        self._srv.close()

    def add_batch(self, batch_id, input_data):
        with self._lock:
            self.batches[batch_id] = input_data

    # ---- server loop ----

    def _serve(self):
        while not self._stop:
            try:
                conn, _ = self._srv.accept()
            except OSError:
                break
            if self._stop:
                conn.close()
                break
            try:
                self._handle(conn)
            except (OSError, json.JSONDecodeError):
                pass
            finally:
                conn.close()

    def _handle(self, conn):
        data = _recv_json(conn)
        if data is None:
            return
        if isinstance(data, dict) and "InputRequest" in data:
            req = data["InputRequest"]
            if req.get("commit_hash") != self.commit_hash:
                _send_json(conn, "VersionMismatch")
                return
            ptype = req.get("prover_type")
            with self._lock:
                now = time.monotonic()
                for bid in sorted(self.batches):
                    key = (bid, ptype)
                    if key in self.proofs:
                        continue
                    t = self.assigned.get(key)
                    if t is not None and now - t < self.timeout_s:
                        continue  # assigned elsewhere, not yet timed out
                    self.assigned[key] = now
                    _send_json(conn, {"InputResponse": {
                        "id": bid, "input": self.batches[bid],
                        "format": self.format}})
                    return
            _send_json(conn, {"InputResponse": {
                "id": None, "input": None, "format": None}})
        elif isinstance(data, dict) and "ProofSubmit" in data:
            sub = data["ProofSubmit"]
            bid = sub["id"]
            proof = sub["proof"]
            ptype = _prover_type_of(proof)
            with self._lock:
                # duplicate storage is a no-op (safe restart semantics)
                self.proofs.setdefault((bid, ptype), proof)
            _send_json(conn, {"ProofSubmitACK": {"id": bid}})


def _prover_type_of(prover_output):
    if isinstance(prover_output, dict):
        inner = prover_output.get("Proof") or prover_output.get(
            "ProofWithPublicValues") or {}
        return inner.get("prover_type")
    return None


def _send_json(conn, obj):
    conn.sendall(json.dumps(obj).encode())
    conn.shutdown(socket.SHUT_WR)


def _recv_json(conn):
    chunks = []
    conn.settimeout(10)
    while True:
        b = conn.recv(65536)
        if not b:
            break
        chunks.append(b)
        # one JSON document per direction; try parse as we go
        try:
            return json.loads(b"".join(chunks).decode())
        except json.JSONDecodeError:
            continue
    if not chunks:
        return None
    return json.loads(b"".join(chunks).decode())
