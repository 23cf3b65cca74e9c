"""Proof-object wire integration: the host-side
mirror of the reference's L1 proof sender
(crates/l2/sequencer/l1_proof_sender.rs:231-563).

What the reference does with finished proofs:
  1. collects CONSECUTIVE proven batches starting after the last verified
     one — a batch is ready only when ALL needed prover types have a proof
     stored for it; the scan stops at the first gap
     (verify_and_send_proofs, l1_proof_sender.rs:298-318);
  2. builds `verifyBatches(uint256,bytes[],bytes[],bytes[])` calldata with
     one proof-bytes array per on-chain-verifiable prover type (RISC0,
     SP1, TDX — Exec has no on-chain verifier), empty bytes where a type
     is not needed (send_verify_batches_tx, l1_proof_sender.rs:514-556);
  3. sends it to the OnChainProposer (or timelock) address.

This module reproduces 1-2 byte-exactly (Solidity ABI encoding included,
selector via the in-repo keccak) and stops at the wire: `calldata` is what
a maintainer hands to their eth client's send_verify_tx.  ProverOutput /
ProofBytes follow crates/common/types/prover.rs:51-100 (the same JSON
shapes ethrex_amd/prover.py emits and the coordinator stores).
"""
from .keccak import selector

VERIFY_BATCHES_SIG = "verifyBatches(uint256,bytes[],bytes[],bytes[])"
# l1_proof_sender.rs:539-545 array order; Exec (0) has no on-chain verifier
ONCHAIN_PROVER_TYPES = ("RISC0", "SP1", "TDX")


def proof_bytes_of(prover_output: dict) -> bytes:
    """ProverOutput::proof_bytes().proof (prover.rs:80-86): both the plain
    Proof and the ProofWithPublicValues variants carry ProofBytes."""
    if "Proof" in prover_output:
        return bytes(prover_output["Proof"]["proof"])
    pb = prover_output["ProofWithPublicValues"]["proof_bytes"]
    return bytes(pb["proof"])


def collect_ready_batches(store: dict, last_verified: int,
                          last_committed: int, needed_types) -> list:
    """Consecutive fully-proven batches after `last_verified`, bounded by
    `last_committed`; stops at the first batch missing any needed type
    (l1_proof_sender.rs:298-318).  `store` maps (batch, type) -> proof."""
    ready = []
    for batch in range(last_verified + 1, last_committed + 1):
        proofs = {}
        for t in needed_types:
            p = store.get((batch, t))
            if p is None:
                return ready  # first gap ends the consecutive run
            proofs[t] = p
        ready.append((batch, proofs))
    return ready


def _abi_bytes(b: bytes) -> bytes:
    pad = (-len(b)) % 32
    return len(b).to_bytes(32, "big") + b + b"\x00" * pad


def _abi_bytes_array(items) -> bytes:
    """dynamic bytes[]: count, per-element offsets, elements"""
    head = len(items).to_bytes(32, "big")
    offsets = b""
    tails = b""
    base = 32 * len(items)
    for it in items:
        offsets += (base + len(tails)).to_bytes(32, "big")
        tails += _abi_bytes(it)
    return head + offsets + tails


def encode_verify_batches_calldata(first_batch: int, batches: list) -> bytes:
    """verifyBatches(uint256,bytes[],bytes[],bytes[]) calldata, one entry
    per batch per array; empty bytes where a prover type has no proof
    (send_verify_batches_tx, l1_proof_sender.rs:514-556)."""
    arrays = []
    for t in ONCHAIN_PROVER_TYPES:
        arrays.append([
            proof_bytes_of(proofs[t]) if t in proofs else b""
            for (_n, proofs) in batches
        ])
    tails = [_abi_bytes_array(a) for a in arrays]
    # head: uint256 + 3 offsets (relative to the start of the args)
    head_len = 4 * 32
    head = first_batch.to_bytes(32, "big")
    off = head_len
    for t in tails:
        head += off.to_bytes(32, "big")
        off += len(t)
    return selector(VERIFY_BATCHES_SIG) + head + b"".join(tails)


def decode_verify_batches_calldata(data: bytes):
    """structural inverse (used by the tests to round-trip the encoder)"""
    assert data[:4] == selector(VERIFY_BATCHES_SIG)
    body = data[4:]
    first_batch = int.from_bytes(body[0:32], "big")
    arrays = []
    for k in range(3):
        off = int.from_bytes(body[32 * (k + 1):32 * (k + 2)], "big")
        cnt = int.from_bytes(body[off:off + 32], "big")
        items = []
        for i in range(cnt):
            eo = off + 32 + int.from_bytes(
                body[off + 32 * (1 + i):off + 32 * (2 + i)], "big")
            ln = int.from_bytes(body[eo:eo + 32], "big")
            items.append(body[eo + 32:eo + 32 + ln])
        arrays.append(items)
    return first_batch, arrays


class ProofSender:
    """verify_and_send mirror over an in-memory rollup store; `send` is a
    callback receiving (target_address, calldata) — the wire boundary."""

    def __init__(self, needed_types, on_chain_proposer, send,
                 timelock_address=None):
        self.needed_types = tuple(needed_types)
        self.target = timelock_address or on_chain_proposer
        self._send = send
        self.store = {}           # (batch, type) -> ProverOutput dict
        self.last_verified = 0
        self.last_committed = 0

    def store_proof(self, batch: int, prover_type: str, output: dict):
        self.store[(batch, prover_type)] = output

    def verify_and_send(self) -> int:
        """sends one verifyBatches tx for the ready run; returns the number
        of batches sent (0 = nothing ready)"""
        ready = collect_ready_batches(self.store, self.last_verified,
                                      self.last_committed, self.needed_types)
        if not ready:
            return 0
        first = ready[0][0]
        calldata = encode_verify_batches_calldata(first, ready)
        self._send(self.target, calldata)
        self.last_verified = ready[-1][0]
        return len(ready)
