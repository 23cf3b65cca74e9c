"""CPU checks of the oracle's BLS12-381 G2 restatement against an
independent pure-Python Fp2/affine implementation (different field
representation, different coordinate system, different algorithm).

Semantics under test = crates/common/crypto/bls_blst.rs:226-255,303-321,
338-345,395-441 (EIP-2537): 192-byte x.c0||x.c1||y.c0||y.c1 points,
canonical coords, (0,0,0,0) identity, on-curve for add, + r-subgroup for
MSM, raw 256-bit scalars.
"""
import random

import pytest

P = 0x1A0111EA397FE69A4B1BA7B6434BACD764774B84F38512BF6730D2A0F6B0F6241EABFFFEB153FFFFB9FEFFFFFFFFAAAB
R = 0x73EDA753299D7D483339D80809A1D80553BDA402FFFE5BFEFFFFFFFF00000001


def f2mul(a, b):
    return ((a[0] * b[0] - a[1] * b[1]) % P, (a[0] * b[1] + a[1] * b[0]) % P)


def f2add(a, b):
    return ((a[0] + b[0]) % P, (a[1] + b[1]) % P)


def f2sub(a, b):
    return ((a[0] - b[0]) % P, (a[1] - b[1]) % P)


def f2inv(a):
    t = pow((a[0] * a[0] + a[1] * a[1]) % P, P - 2, P)
    return (a[0] * t % P, (-a[1]) * t % P)


def g2add(p, q):
    if p is None:
        return q
    if q is None:
        return p
    if p[0] == q[0]:
        if f2add(p[1], q[1]) == (0, 0):
            return None
        lam = f2mul(f2mul((3, 0), f2mul(p[0], p[0])),
                    f2inv(f2mul((2, 0), p[1])))
    else:
        lam = f2mul(f2sub(q[1], p[1]), f2inv(f2sub(q[0], p[0])))
    x3 = f2sub(f2sub(f2mul(lam, lam), p[0]), q[0])
    return (x3, f2sub(f2mul(lam, f2sub(p[0], x3)), p[1]))


def g2mul(k, p):
    r = None
    while k:
        if k & 1:
            r = g2add(r, p)
        p = g2add(p, p)
        k >>= 1
    return r


def dec(b):
    c = [int.from_bytes(b[48 * i:48 * i + 48], "big") for i in range(4)]
    if all(v == 0 for v in c):
        return None
    return ((c[0], c[1]), (c[2], c[3]))


def enc(p):
    if p is None:
        return b"\x00" * 192
    return b"".join(v.to_bytes(48, "big")
                    for v in (p[0][0], p[0][1], p[1][0], p[1][1]))


@pytest.fixture(scope="module")
def gen5(oracle_mod):
    return oracle_mod.bls_g2_gen_points(0, 5)


def test_g2_gen_points_on_curve_in_subgroup(oracle_mod, gen5):
    g = dec(gen5[:192])
    for i in range(5):
        q = dec(gen5[192 * i:192 * (i + 1)])
        lhs = f2mul(q[1], q[1])
        rhs = f2add(f2mul(f2mul(q[0], q[0]), q[0]), (4, 4))
        assert lhs == rhs, f"off curve {i}"
        assert g2mul(R, q) is None, f"not in subgroup {i}"
        assert q == g2mul(i + 1, g), f"chain {i}"


def test_g2_add_parity(oracle_mod, gen5):
    g = dec(gen5[:192])
    rc, s = oracle_mod.bls_g2_add(gen5[:192], gen5[192:384])
    assert rc == 0 and dec(s) == g2mul(3, g)
    rc, d = oracle_mod.bls_g2_add(gen5[:192], gen5[:192])  # doubling
    assert rc == 0 and dec(d) == g2mul(2, g)
    neg = enc((g[0], f2sub((0, 0), g[1])))
    rc, z = oracle_mod.bls_g2_add(gen5[:192], neg)  # P + (-P)
    assert rc == 0 and dec(z) is None
    rc, i1 = oracle_mod.bls_g2_add(gen5[:192], b"\x00" * 192)  # + identity
    assert rc == 0 and dec(i1) == g


def test_g2_mul_parity(oracle_mod, gen5):
    g = dec(gen5[:192])
    for k in (0, 1, 2, R - 1, R, R + 5, (1 << 256) - 1):
        rc, m = oracle_mod.bls_g2_mul(gen5[:192], k.to_bytes(32, "big"))
        assert rc == 0 and dec(m) == g2mul(k % R, g), hex(k)


def test_g2_msm_parity(oracle_mod):
    n = 8
    pts = oracle_mod.bls_g2_gen_points(3, n)
    random.seed(9)
    scs = b"".join(random.randrange(1 << 255).to_bytes(32, "big")
                   for _ in range(n))
    rc, got = oracle_mod.bls_g2_msm(pts, scs, n)
    assert rc == 0
    want = None
    for i in range(n):
        k = int.from_bytes(scs[32 * i:32 * i + 32], "big") % R
        want = g2add(want, g2mul(k, dec(pts[192 * i:192 * (i + 1)])))
    assert dec(got) == want


def _non_subgroup_point():
    """an on-curve G2 point OUTSIDE the r-subgroup (cofactor not cleared)"""
    def f2pow(a, e):
        r = (1, 0)
        while e:
            if e & 1:
                r = f2mul(r, a)
            a = f2mul(a, a)
            e >>= 1
        return r
    for ctr in range(1, 64):
        x = (ctr, 1)
        rhs = f2add(f2mul(f2mul(x, x), x), (4, 4))
        a1 = f2pow(rhs, (P - 3) // 4)
        x0 = f2mul(a1, rhs)
        alpha = f2mul(a1, x0)
        if alpha == (P - 1, 0):
            y = f2mul((0, 1), x0)
        else:
            y = f2mul(f2pow(f2add(alpha, (1, 0)), (P - 1) // 2), x0)
        if f2mul(y, y) != rhs:
            continue
        if g2mul(R, (x, y)) is not None:
            return (x, y)
    raise AssertionError("no non-subgroup point found")


def test_g2_msm_rejects_non_subgroup(oracle_mod):
    q = _non_subgroup_point()
    rc, _ = oracle_mod.bls_g2_msm(enc(q), (5).to_bytes(32, "big"), 1)
    assert rc == 3  # BORC_ERR_SUBGROUP
    # add does NOT subgroup-check (EIP-2537 G2ADD)
    rc, _ = oracle_mod.bls_g2_add(enc(q), enc(q))
    assert rc == 0


def test_g2_rejects_bad_input(oracle_mod, gen5):
    bad = bytearray(gen5[:192])
    bad[191] ^= 1
    rc, _ = oracle_mod.bls_g2_add(bytes(bad), gen5[:192])
    assert rc == 1  # off curve
    noncanon = P.to_bytes(48, "big") + b"\x00" * 144
    rc, _ = oracle_mod.bls_g2_add(noncanon, gen5[:192])
    assert rc == 2  # non-canonical coordinate
