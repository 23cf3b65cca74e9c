"""Test-infrastructure reference MPT: direct recursive root computation
(hash-as-you-go), independent of the product's level-synchronized builder
(ethrex_amd/trie.py).  Used only by tests."""
from ethrex_amd import rlp
from ethrex_amd.keccak import keccak256
from ethrex_amd.trie import _hex_prefix, _nibbles  # shared canonical codec
from ethrex_amd.witness import EMPTY_TRIE_HASH


def _node(items):
    """items: sorted (nibble-list, value) pairs -> RLP encoding of node."""
    if len(items) == 1:
        nib, val = items[0]
        return rlp.encode([_hex_prefix(nib, True), val])
    first, last = items[0][0], items[-1][0]
    lcp = 0
    while lcp < len(first) and lcp < len(last) and first[lcp] == last[lcp]:
        lcp += 1
    if lcp:
        child = _node([(n[lcp:], v) for n, v in items])
        return rlp.encode([_hex_prefix(first[:lcp], False), _ref(child)])
    slots = [b""] * 16
    value = b""
    i = 0
    while i < len(items):
        nib, val = items[i]
        if not nib:
            value = val
            i += 1
            continue
        j = i
        while j < len(items) and items[j][0] and items[j][0][0] == nib[0]:
            j += 1
        slots[nib[0]] = _ref(_node([(n[1:], v) for n, v in items[i:j]]))
        i = j
    return rlp.encode(slots + [value])


def _ref(encoded: bytes):
    if len(encoded) >= 32:
        return keccak256(encoded)
    return rlp.decode(encoded)


def mpt_root(pairs: dict) -> bytes:
    if not pairs:
        return EMPTY_TRIE_HASH
    items = sorted((_nibbles(k), v) for k, v in pairs.items())
    return keccak256(_node(items))
