"""Size-independent algebraic properties of the oracle (tier rule ③):
round trips, linearity, MSM decomposition identities — these are the checks
that scale up to BASELINE sizes where direct vectors are impractical.
"""
import pytest

R = 0x30644E72E131A029B85045B68181585D2833E84879B9709143E1F593F0000001


def test_msm_pippenger_vs_naive(oracle_mod):
    for n in (1, 2, 3, 31, 200):
        pts = oracle_mod.gen_points(0, n)
        scs = oracle_mod.gen_fr(42, n)
        rc1, a = oracle_mod.g1_msm(pts, scs, n)
        rc2, b = oracle_mod.g1_msm_naive(pts, scs, n)
        assert rc1 == rc2 == 0 and a == b, n


def test_msm_basis_identity(oracle_mod):
    """MSM over unit-like inputs: scalars = e_i picks out k*P_i."""
    n = 8
    pts = oracle_mod.gen_points(0, n)
    for i in (0, 3, 7):
        scs = bytearray(32 * n)
        scs[32 * i:32 * (i + 1)] = (1234567).to_bytes(32, "big")
        rc, out = oracle_mod.g1_msm(pts, bytes(scs), n)
        rc2, exp = oracle_mod.g1_mul(pts[64 * i:64 * (i + 1)],
                                     (1234567).to_bytes(32, "big"))
        assert rc == rc2 == 0 and out == exp


def test_msm_shard_combine(oracle_mod):
    """Sharded partial sums combine to the unsharded result — the identity
    the 8-GPU RCCL exchange relies on."""
    n = 96
    pts = oracle_mod.gen_points(0, n)
    scs = oracle_mod.gen_fr(42, n)
    rc, full = oracle_mod.g1_msm(pts, scs, n)
    assert rc == 0
    parts = b""
    for g in range(4):
        lo, hi = g * 24, (g + 1) * 24
        rc, j = oracle_mod.g1_msm_jacobian(pts[64 * lo:64 * hi],
                                           scs[32 * lo:32 * hi], 24)
        assert rc == 0
        parts += j
    rc, combined = oracle_mod.g1_combine_jacobian(parts, 4)
    assert rc == 0 and combined == full


def test_ntt_roundtrip_medium(oracle_mod):
    for n in (64, 1024, 4096):
        a = oracle_mod.gen_fr(43, n)
        rc, f = oracle_mod.fr_ntt(a, n, False)
        assert rc == 0
        rc, back = oracle_mod.fr_ntt(f, n, True)
        assert rc == 0 and back == a, n


def test_ntt_linearity(oracle_mod):
    """NTT(a + c*b) == NTT(a) + c*NTT(b) elementwise."""
    n = 64
    a_b = oracle_mod.gen_fr(43, 2 * n)
    a = [int.from_bytes(a_b[32 * i:32 * i + 32], "big") for i in range(n)]
    b = [int.from_bytes(a_b[32 * (n + i):32 * (n + i) + 32], "big") for i in range(n)]
    c = 987654321
    comb = b"".join(((a[i] + c * b[i]) % R).to_bytes(32, "big") for i in range(n))
    rc, fa = oracle_mod.fr_ntt(b"".join(x.to_bytes(32, "big") for x in a), n, False)
    rc2, fb = oracle_mod.fr_ntt(b"".join(x.to_bytes(32, "big") for x in b), n, False)
    rc3, fc = oracle_mod.fr_ntt(comb, n, False)
    assert rc == rc2 == rc3 == 0
    for j in range(n):
        fa_j = int.from_bytes(fa[32 * j:32 * j + 32], "big")
        fb_j = int.from_bytes(fb[32 * j:32 * j + 32], "big")
        fc_j = int.from_bytes(fc[32 * j:32 * j + 32], "big")
        assert fc_j == (fa_j + c * fb_j) % R


def test_ntt_delta_is_constant(oracle_mod):
    """NTT of delta at 0 is all-ones; NTT of constant c is (n*c, 0, ...)."""
    n = 32
    delta = (1).to_bytes(32, "big") + b"\x00" * (32 * (n - 1))
    rc, f = oracle_mod.fr_ntt(delta, n, False)
    assert rc == 0
    one = (1).to_bytes(32, "big")
    assert f == one * n
    const = (7).to_bytes(32, "big") * n
    rc, f = oracle_mod.fr_ntt(const, n, False)
    assert rc == 0
    assert f[:32] == (7 * n).to_bytes(32, "big")
    assert f[32:] == b"\x00" * (32 * (n - 1))


def test_gen_fr_deterministic_and_distinct(oracle_mod):
    a = oracle_mod.gen_fr(42, 100)
    b = oracle_mod.gen_fr(42, 100)
    c = oracle_mod.gen_fr(43, 100)
    assert a == b and a != c
    for i in range(100):
        assert int.from_bytes(a[32 * i:32 * i + 32], "big") < R
