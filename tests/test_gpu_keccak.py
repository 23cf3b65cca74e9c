"""GPU parity for batched keccak256 (witness/statement
hashing — the reference hashes per-node through hand-written asm,
crates/common/crypto/keccak/mod.rs).  Oracle: ethrex_amd/keccak.py, pinned
by the canonical Keccak-256 vectors.
"""
import random

import pytest

from ethrex_amd.keccak import keccak256

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    ethrex_amd.set_device(0)
    return ethrex_amd


def _batch(lengths, seed=0):
    random.seed(seed)
    msgs = b"".join(bytes(random.randrange(256) for _ in range(ln))
                    for ln in lengths)
    offs = [0]
    for ln in lengths:
        offs.append(offs[-1] + ln)
    return msgs, offs


def test_gpu_keccak_rate_boundaries(gpu):
    """empty, 1 byte, and every boundary around the 136-byte rate"""
    lengths = [0, 1, 31, 32, 33, 64, 134, 135, 136, 137, 200,
               271, 272, 273, 544, 545]
    msgs, offs = _batch(lengths, seed=3)
    rc, out = gpu.keccak256_batch(msgs, offs, len(lengths))
    assert rc == 0
    for i, ln in enumerate(lengths):
        want = keccak256(msgs[offs[i]:offs[i + 1]])
        assert out[32 * i:32 * i + 32] == want, f"len={ln}"


def test_gpu_keccak_canonical_vectors(gpu):
    msgs = b"abc"
    rc, out = gpu.keccak256_batch(msgs, [0, 0, 3], 2)
    assert rc == 0
    assert out[:32].hex() == ("c5d2460186f7233c927e7db2dcc703c0"
                              "e500b653ca82273b7bfad8045d85a470")
    assert out[32:].hex() == ("4e03657aea45a94fc7d47ba826c8d667"
                              "c0d1e6e33a64a036ec44f58fa12d6c45")


def test_gpu_keccak_trie_shaped_batch(gpu):
    """512 messages with trie-node-like lengths (32..532 B), exact parity"""
    random.seed(11)
    lengths = [random.choice([32, 64, 83, 104, 532, 17, 200])
               for _ in range(512)]
    msgs, offs = _batch(lengths, seed=12)
    rc, out = gpu.keccak256_batch(msgs, offs, len(lengths))
    assert rc == 0
    for i in random.sample(range(len(lengths)), 64):
        assert out[32 * i:32 * i + 32] == keccak256(msgs[offs[i]:offs[i + 1]])


def test_gpu_keccak_plan_large_batch(gpu):
    """2^18 x 136-byte messages through the resident plan; determinism +
    spot parity (the pure-Python oracle is too slow for full coverage)"""
    n = 1 << 18
    blob = bytes((i * 131 + 7) & 0xFF for i in range(136 * 64))
    msgs = (blob * ((n // 64) + 1))[:136 * n]
    offs = list(range(0, 136 * (n + 1), 136))
    plan = gpu.KeccakPlan(len(msgs), n)
    plan.upload(msgs, offs)
    plan.run()
    out1 = plan.download()
    plan.run()
    assert plan.download() == out1  # deterministic
    plan.destroy()
    random.seed(13)
    for i in random.sample(range(n), 32):
        assert out1[32 * i:32 * i + 32] == \
            keccak256(msgs[136 * i:136 * (i + 1)])
