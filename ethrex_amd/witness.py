"""Witness validation / statement generation — the stateless-validation
half of `ExecBackend::execute_core` this core accelerates.

Restates the reference's witness-to-tries rebuild
(crates/common/types/block_execution_witness.rs:
 `RpcExecutionWitness::into_execution_witness` /
 `rebuild_state_and_storage_tries`, lines ~200-380):

  1. keccak-256 EVERY witness state node (the bulk hashing work — here a
     single batched GPU launch through the caller-supplied `hash_batch`;
     the reference hashes per node through its `crypto.keccak256`);
     undecodable entries and the 0x80 Null sentinel are skipped, per EELS
     `test_validation_state_extra_unused_trie_node` semantics;
  2. recover the initial state root from the parent header of the first
     block (`find_parent_state_root`);
  3. link the state trie from that root through the hash->node map,
     collect accounts and their storage roots from the leaves, and link
     every storage trie whose root node is present (absent roots and
     EMPTY_TRIE_HASH are skipped, witness.rs:380-392).

The STATEMENT binds the proof to the witness: the initial state root,
the linked-trie shape counts, and a keccak commitment over them.  (Full
EVM re-execution — final_state_hash — is the out-of-scope zkVM guest
half; see DESIGN.md.)
"""
from . import rlp
from .keccak import keccak256

# keccak256(rlp(b'')) = keccak256(0x80) — pinned in the reference,
# crates/common/trie/trie.rs:78 (EMPTY_TRIE_HASH)
EMPTY_TRIE_HASH = bytes.fromhex(
    "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")

# BlockHeader RLP field positions (crates/common/types/block.rs header RLP)
_H_STATE_ROOT = 3
_H_NUMBER = 8


def cpu_hash_batch(msgs):
    """Host keccak batch — the ExecBackend (reference exec.rs CPU path)
    and CPU tests; the product Mi355 path passes the GPU KeccakPlan."""
    return [keccak256(m) for m in msgs]


def find_parent_state_root(headers_rlp, first_block_number: int) -> bytes:
    """find_parent_state_root (block_execution_witness.rs): the state root
    of the header whose number == first_block_number - 1."""
    for h in headers_rlp:
        fields = rlp.decode(h)
        number = int.from_bytes(fields[_H_NUMBER], "big")
        if number == first_block_number - 1:
            return fields[_H_STATE_ROOT]
    raise ValueError("witness: parent header not found")


def _node_kind(item):
    """17-item -> branch; 2-item -> ext/leaf by the hex-prefix flag."""
    if len(item) == 17:
        return "branch"
    if len(item) == 2 and item[0]:
        flag = item[0][0] >> 4
        return "leaf" if flag >= 2 else "ext"
    raise ValueError("witness: malformed trie node")


def _children(item):
    """child references of a decoded node: (ref, is_value_leaf) pairs.
    A ref is 32-byte bytes (hash) or a decoded inline node (list)."""
    kind = _node_kind(item)
    if kind == "branch":
        return [c for c in item[:16] if c != b""], item[16]
    if kind == "ext":
        return [item[1]], b""
    return [], item[1]  # leaf: value only


class LinkedTrie:
    """Result of linking one trie through the node map."""

    def __init__(self):
        self.n_nodes = 0
        self.leaf_values = []
        self.missing = 0  # hash refs not present in the witness (allowed:
        #                   the witness carries only the accessed subtrie)


def link_trie(root_hash: bytes, node_map: dict) -> LinkedTrie:
    """Walk a trie from `root_hash` over the hash->decoded-node map
    (Trie::get_embedded_root_committed shape: children resolve by hash;
    inline (<32 B) nodes are embedded directly)."""
    out = LinkedTrie()
    if root_hash == EMPTY_TRIE_HASH:
        return out
    if root_hash not in node_map:
        raise ValueError("witness: RootNotFound")  # witness.rs root check
    stack = [node_map[root_hash]]
    out.n_nodes = 1
    while stack:
        item = stack.pop()
        kids, value = _children(item)
        if value:
            out.leaf_values.append(value)
        for ref in kids:
            if isinstance(ref, list):  # inline node (< 32 B encoding)
                out.n_nodes += 1
                stack.append(ref)
            elif ref in node_map:
                out.n_nodes += 1
                stack.append(node_map[ref])
            else:
                out.missing += 1
    return out


def _path_nibbles(hp: bytes):
    """decode a hex-prefix path -> (nibble list, is_leaf)"""
    flag = hp[0] >> 4
    nibs = [] if flag in (0, 2) else [hp[0] & 0xF]
    for b in hp[1:]:
        nibs.append(b >> 4)
        nibs.append(b & 0xF)
    return nibs, flag >= 2


def collect_leaves(root_hash: bytes, node_map: dict):
    """Walk a FULLY-present trie collecting {key bytes: value}; returns
    (leaves, missing_count).  Paths are rebuilt from branch indices +
    hex-prefix segments (nibbles.rs / node path semantics)."""
    leaves = {}
    missing = 0
    if root_hash == EMPTY_TRIE_HASH:
        return leaves, missing
    stack = [(node_map[root_hash], [])]
    while stack:
        item, path = stack.pop()
        kind = _node_kind(item)
        if kind == "branch":
            if item[16]:
                leaves[_nib_bytes(path)] = item[16]
            for i in range(16):
                ref = item[i]
                if ref == b"":
                    continue
                if isinstance(ref, list):
                    stack.append((ref, path + [i]))
                elif ref in node_map:
                    stack.append((node_map[ref], path + [i]))
                else:
                    missing += 1
        elif kind == "ext":
            nibs, _ = _path_nibbles(item[0])
            ref = item[1]
            if isinstance(ref, list):
                stack.append((ref, path + nibs))
            elif ref in node_map:
                stack.append((node_map[ref], path + nibs))
            else:
                missing += 1
        else:
            nibs, _ = _path_nibbles(item[0])
            leaves[_nib_bytes(path + nibs)] = item[1]
    return leaves, missing


def _nib_bytes(nibs):
    assert len(nibs) % 2 == 0
    return bytes(nibs[i] << 4 | nibs[i + 1] for i in range(0, len(nibs), 2))


def build_node_map(state_nodes, hash_batch):
    """hash->decoded-node map over the witness nodes (sentinel and
    undecodable entries skipped); returns (node_map, skipped)."""
    hashes = hash_batch(list(state_nodes))
    node_map = {}
    skipped = 0
    for b, h in zip(state_nodes, hashes):
        if b == b"\x80":
            skipped += 1
            continue
        try:
            item = rlp.decode(b)
            if not isinstance(item, list):
                raise ValueError
            _node_kind(item)
        except ValueError:
            skipped += 1
            continue
        node_map[h] = item
    return node_map, skipped


def validate_witness(state_nodes, headers_rlp, first_block_number: int,
                     hash_batch) -> dict:
    """The witness-validation statement.  `state_nodes` = raw RLP node
    bytes (witness.state); `hash_batch(list[bytes]) -> list[32-byte]` is
    the keccak engine (GPU KeccakPlan on the product path)."""
    node_map, skipped = build_node_map(state_nodes, hash_batch)
    initial_state_root = find_parent_state_root(headers_rlp,
                                                first_block_number)
    state = link_trie(initial_state_root, node_map)
    # accounts: state-trie leaf values are RLP([nonce, balance,
    # storage_root, code_hash]); link each present storage trie
    n_accounts = 0
    n_storage_tries = 0
    storage_nodes = 0
    for v in state.leaf_values:
        acct = rlp.decode(v)
        if not (isinstance(acct, list) and len(acct) == 4):
            continue
        n_accounts += 1
        sroot = acct[2]
        if sroot == EMPTY_TRIE_HASH or sroot not in node_map:
            continue  # witness.rs:380-392: skip absent storage roots
        st = link_trie(sroot, node_map)
        n_storage_tries += 1
        storage_nodes += st.n_nodes
    commitment = keccak256(
        initial_state_root
        + len(state_nodes).to_bytes(8, "big")
        + state.n_nodes.to_bytes(8, "big")
        + n_accounts.to_bytes(8, "big")
        + n_storage_tries.to_bytes(8, "big")
        + storage_nodes.to_bytes(8, "big"))
    return {
        "initial_state_root": initial_state_root.hex(),
        "n_witness_nodes": len(state_nodes),
        "n_skipped": skipped,
        "n_state_nodes_linked": state.n_nodes,
        "n_accounts": n_accounts,
        "n_storage_tries": n_storage_tries,
        "n_storage_nodes_linked": storage_nodes,
        "commitment": commitment.hex(),
    }


def load_witness_fixture(path):
    """Load the committed hoodi witness fixture (tests/golden/*.json.gz,
    extracted from the reference's fixtures/cache/rpc_prover cache)."""
    import gzip
    import json
    with gzip.open(path, "rt") as f:
        d = json.load(f)
    state = [bytes.fromhex(s[2:]) for s in d["state"]]
    headers = [bytes.fromhex(s[2:]) for s in d["headers"]]
    return state, headers, d["first_block_number"]
