"""Pin the CPU oracle against the committed golden fixtures.

Fixture provenance: tests/golden/gen_fixtures.py — the reference's in-tree
vectors (crypto_parity.rs:26-119) restated + pure-Python bignum expected
values.  This is the step that makes the oracle trustworthy as the parity
anchor for the HIP kernels (tier rule ③).
"""
import pytest


def _b(h):
    return bytes.fromhex(h)


def test_g1_add_vectors(golden, oracle_mod):
    for v in golden["g1_add"]:
        rc, out = oracle_mod.g1_add(_b(v["a"]), _b(v["b"]))
        if v.get("error"):
            assert rc != 0, v["name"]
        else:
            assert rc == 0 and out == _b(v["out"]), v["name"]


def test_g1_mul_vectors(golden, oracle_mod):
    for v in golden["g1_mul"]:
        rc, out = oracle_mod.g1_mul(_b(v["point"]), _b(v["scalar"]))
        assert rc == 0 and out == _b(v["out"]), v["name"]


def test_fq_mul_vectors(golden, oracle_mod):
    for v in golden["fq_mul"]:
        assert oracle_mod.fq_mulmod(_b(v["a"]), _b(v["b"])) == _b(v["out"])


def test_fr_mul_vectors(golden, oracle_mod):
    for v in golden["fr_mul"]:
        assert oracle_mod.fr_mulmod(_b(v["a"]), _b(v["b"])) == _b(v["out"])


def test_msm_vectors(golden, oracle_mod):
    for v in golden["msm"]:
        rc, out = oracle_mod.g1_msm(_b(v["points"]), _b(v["scalars"]), v["n"])
        assert rc == 0 and out == _b(v["out"]), f"msm n={v['n']}"
        rc, out = oracle_mod.g1_msm_naive(_b(v["points"]), _b(v["scalars"]), v["n"])
        assert rc == 0 and out == _b(v["out"]), f"naive msm n={v['n']}"


def test_ntt_vectors(golden, oracle_mod):
    for v in golden["ntt"]:
        rc, fwd = oracle_mod.fr_ntt(_b(v["in"]), v["n"], False)
        assert rc == 0 and fwd == _b(v["fwd"]), f"ntt n={v['n']}"
        rc, back = oracle_mod.fr_ntt(fwd, v["n"], True)
        assert rc == 0 and back == _b(v["inv_of_fwd"]), f"intt n={v['n']}"


def test_ntt_rejects_noncanonical(oracle_mod):
    rc, _ = oracle_mod.fr_ntt(b"\xff" * 32, 1, False)
    assert rc != 0


def test_ntt_rejects_nonpow2(oracle_mod):
    rc, _ = oracle_mod.fr_ntt(b"\x00" * 96, 3, False)
    assert rc != 0
