"""Minimal RLP codec (the subset Merkle-Patricia-trie nodes and block
headers use), restated from the reference's ethrex-rlp semantics
(crates/common/rlp/src/{encode,decode}.rs; canonical Ethereum RLP).

Items are bytes or (recursively) lists of items.
"""


def encode(item) -> bytes:
    if isinstance(item, (bytes, bytearray)):
        b = bytes(item)
        if len(b) == 1 and b[0] < 0x80:
            return b
        return _len_prefix(len(b), 0x80) + b
    if isinstance(item, list):
        payload = b"".join(encode(x) for x in item)
        return _len_prefix(len(payload), 0xC0) + payload
    raise TypeError(f"rlp: cannot encode {type(item)}")


def _len_prefix(n: int, base: int) -> bytes:
    if n < 56:
        return bytes([base + n])
    nb = n.to_bytes((n.bit_length() + 7) // 8, "big")
    return bytes([base + 55 + len(nb)]) + nb


def decode(b: bytes):
    """Decode one top-level item; rejects trailing bytes."""
    item, end = _decode_at(b, 0)
    if end != len(b):
        raise ValueError("rlp: trailing bytes")
    return item


def _decode_at(b: bytes, i: int):
    if i >= len(b):
        raise ValueError("rlp: truncated")
    p = b[i]
    if p < 0x80:
        return b[i:i + 1], i + 1
    if p < 0xB8:
        n = p - 0x80
        if i + 1 + n > len(b):
            raise ValueError("rlp: truncated string")
        if n == 1 and b[i + 1] < 0x80:
            raise ValueError("rlp: non-canonical single byte")
        return b[i + 1:i + 1 + n], i + 1 + n
    if p < 0xC0:
        ln = p - 0xB7
        n = int.from_bytes(b[i + 1:i + 1 + ln], "big")
        if ln == 0 or b[i + 1] == 0 or n < 56:
            raise ValueError("rlp: non-canonical length")
        s = i + 1 + ln
        if s + n > len(b):
            raise ValueError("rlp: truncated string")
        return b[s:s + n], s + n
    if p < 0xF8:
        n = p - 0xC0
        end = i + 1 + n
    else:
        ln = p - 0xF7
        n = int.from_bytes(b[i + 1:i + 1 + ln], "big")
        if ln == 0 or b[i + 1] == 0 or n < 56:
            raise ValueError("rlp: non-canonical length")
        end = i + 1 + ln + n
    if end > len(b):
        raise ValueError("rlp: truncated list")
    items = []
    j = i + 1 if p < 0xF8 else i + 1 + (p - 0xF7)
    while j < end:
        it, j = _decode_at(b, j)
        items.append(it)
    if j != end:
        raise ValueError("rlp: list length mismatch")
    return items, end
