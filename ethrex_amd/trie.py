"""Merkle-Patricia-trie root computation with LEVEL-SYNCHRONIZED batched
node hashing (second half: the reference hashes per
trie node through crates/common/trie — Trie::compute_hash walks the tree
hashing each node's RLP; here every tree level is hashed as ONE batched
keccak launch, the GPU KeccakPlan on the product path).

Node encoding restated from crates/common/trie/node/{branch,extension,
leaf}.rs + node_hash.rs:
  branch = RLP 17-list: 16 child refs + value
  ext    = RLP 2-list: hex-prefix(path, flag 0/1) + child ref
  leaf   = RLP 2-list: hex-prefix(path, flag 2/3) + value
  ref    = keccak256(rlp) if len(rlp) >= 32 else the rlp itself (inline)
Pinned against the reference by re-encoding the hoodi witness fixture's
1705 real nodes byte-exactly (tests/test_witness.py) and by the
EMPTY_TRIE_HASH constant (trie.rs:78).
"""
from . import rlp
from .witness import EMPTY_TRIE_HASH


def _nibbles(key: bytes):
    out = []
    for b in key:
        out.append(b >> 4)
        out.append(b & 0xF)
    return out


def _hex_prefix(nibbles, leaf: bool) -> bytes:
    flag = 2 if leaf else 0
    if len(nibbles) % 2:
        head = [(flag + 1) << 4 | nibbles[0]]
        rest = nibbles[1:]
    else:
        head = [flag << 4]
        rest = nibbles
    body = bytes(head) + bytes(
        rest[i] << 4 | rest[i + 1] for i in range(0, len(rest), 2))
    return body


class _Node:
    __slots__ = ("kind", "path", "children", "value", "depth", "encoded",
                 "ref")

    def __init__(self, kind):
        self.kind = kind          # "branch" | "ext" | "leaf"
        self.path = []
        self.children = {}        # branch: nibble -> _Node; ext: {0: child}
        self.value = b""
        self.depth = 0
        self.encoded = None
        self.ref = None           # bytes32 hash or inline rlp (< 32 B)


def _build(items):
    """Build the radix structure for sorted (nibble-list, value) pairs."""
    if not items:
        return None
    if len(items) == 1:
        nib, val = items[0]
        leaf = _Node("leaf")
        leaf.path = nib
        leaf.value = val
        return leaf
    # longest common prefix
    first, last = items[0][0], items[-1][0]
    lcp = 0
    while (lcp < len(first) and lcp < len(last)
           and first[lcp] == last[lcp]):
        lcp += 1
    if lcp > 0:
        ext = _Node("ext")
        ext.path = first[:lcp]
        ext.children[0] = _build([(n[lcp:], v) for n, v in items])
        return ext
    branch = _Node("branch")
    i = 0
    while i < len(items):
        nib, val = items[i]
        if not nib:  # value lands at this branch
            branch.value = val
            i += 1
            continue
        j = i
        while j < len(items) and items[j][0] and items[j][0][0] == nib[0]:
            j += 1
        branch.children[nib[0]] = _build(
            [(n[1:], v) for n, v in items[i:j]])
        i = j
    return branch


def trie_root(pairs, hash_batch) -> bytes:
    """Root hash of the MPT over {key bytes: value bytes}, hashing each
    tree level as one `hash_batch` call (deepest level first — children
    resolve before parents encode).  Same root as the reference's
    Trie::compute_hash for the same pairs."""
    if not pairs:
        return EMPTY_TRIE_HASH
    items = sorted((_nibbles(k), v) for k, v in pairs.items())
    root = _build(items)
    # collect nodes by depth
    levels = []

    def visit(node, d):
        node.depth = d
        while len(levels) <= d:
            levels.append([])
        levels[d].append(node)
        for c in node.children.values():
            visit(c, d + 1)

    visit(root, 0)
    # deepest level first: encode (children already have refs), batch-hash
    for level in reversed(levels):
        to_hash = []
        for n in level:
            if n.kind == "leaf":
                enc = rlp.encode([_hex_prefix(n.path, True), n.value])
            elif n.kind == "ext":
                enc = rlp.encode([_hex_prefix(n.path, False),
                                  _ref_item(n.children[0])])
            else:
                enc = rlp.encode(
                    [_ref_item(n.children[i]) if i in n.children else b""
                     for i in range(16)] + [n.value])
            n.encoded = enc
            if len(enc) >= 32:
                to_hash.append(n)
        if to_hash:
            hashes = hash_batch([n.encoded for n in to_hash])
            for n, h in zip(to_hash, hashes):
                n.ref = h
        for n in level:
            if n.ref is None:
                n.ref = n.encoded  # inline (< 32 B)
    # the root is ALWAYS hashed, even if < 32 B (node_hash.rs finalize)
    if len(root.encoded) < 32:
        return hash_batch([root.encoded])[0]
    return root.ref


def _ref_item(child):
    """RLP item for a child reference: 32-byte string for hashed children,
    the decoded inline node for < 32 B encodings (the raw rlp embeds)."""
    if len(child.encoded) < 32:
        return rlp.decode(child.encoded)
    return child.ref


def account_leaf(nonce: int, balance: int, storage_root: bytes,
                 code_hash: bytes) -> bytes:
    """RLP([nonce, balance, storage_root, code_hash]) — the state-trie
    leaf value (AccountState RLP, crates/common/types/account.rs)."""
    def _int(v):
        if v == 0:
            return b""
        return v.to_bytes((v.bit_length() + 7) // 8, "big")
    return rlp.encode([_int(nonce), _int(balance), storage_root, code_hash])


def trie_root_hashed_keys(pairs, hash_batch, hash_packed=None) -> bytes:
    """MPT root over FIXED 32-byte keys through the NATIVE structure
    builder (C host; ~100x the Python radix build) with the same
    level-synchronized hashing: every level is one `hash_batch` call
    (the GPU KeccakPlan on the product path).  Byte-identical to
    trie_root() — pinned by tests on random sets and the hoodi witness
    storage tries."""
    from .lib import MptBuilder
    if not pairs:
        return EMPTY_TRIE_HASH
    keys = sorted(pairs)
    assert all(len(k) == 32 for k in keys)
    vals = b"".join(pairs[k] for k in keys)
    offs = [0]
    for k in keys:
        offs.append(offs[-1] + len(pairs[k]))
    b = MptBuilder(b"".join(keys), vals, offs)
    try:
        for d in range(b.max_depth(), -1, -1):
            msgs, moffs = b.level_encode(d)
            if len(moffs) > 1:
                # hash_packed takes the level buffer in the batched-keccak
                # layout directly (no per-message slicing; the bench reuses
                # one device KeccakPlan across levels)
                if hash_packed is not None:
                    hashes = hash_packed(msgs, moffs)
                else:
                    hashes = hash_batch_encoded(msgs, moffs, hash_batch)
                b.level_set_hashes(d, hashes)
            else:
                b.level_set_hashes(d, b"")
        return b.root()
    finally:
        b.destroy()


def hash_batch_encoded(msgs: bytes, offs, hash_batch) -> bytes:
    """adapt a list-of-bytes hash_batch to the packed (buf, offsets) level
    layout the native builder emits"""
    parts = [msgs[offs[i]:offs[i + 1]] for i in range(len(offs) - 1)]
    return b"".join(hash_batch(parts))
