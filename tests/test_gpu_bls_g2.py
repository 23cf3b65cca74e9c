"""GPU parity for BLS12-381 G2 (the G2 half of the
EIP-2537 precompile surface): the gfx950 Fp2/G2 path through the C-ABI vs
the CPU oracle, bit-exact.  Semantics: bls_blst.rs:338-345 (g2_add, no
subgroup check) and :395-441 (g2_msm, subgroup-checked, raw scalars).
"""
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    ethrex_amd.set_device(0)
    return ethrex_amd


def test_gpu_g2_gen_points_parity(gpu, oracle_mod):
    n = 64
    plan = gpu.BlsG2MsmPlan(n)
    plan.gen_points(0)
    got = plan.download_points()
    plan.destroy()
    assert got == oracle_mod.bls_g2_gen_points(0, n)


def test_gpu_g2_add_mul_parity(gpu, oracle_mod):
    pts = oracle_mod.bls_g2_gen_points(0, 3)
    a, b = pts[:192], pts[192:384]
    rc, got = gpu.bls_g2_add(a, b)
    rc2, want = oracle_mod.bls_g2_add(a, b)
    assert rc == rc2 == 0 and got == want
    rc, got = gpu.bls_g2_add(a, a)  # doubling branch
    rc2, want = oracle_mod.bls_g2_add(a, a)
    assert rc == rc2 == 0 and got == want
    rc, got = gpu.bls_g2_add(a, b"\x00" * 192)  # identity operand
    assert rc == 0 and got == a
    for k in (0, 1, 5, (1 << 256) - 1):
        kb = k.to_bytes(32, "big")
        rc, got = gpu.bls_g2_mul(a, kb)
        rc2, want = oracle_mod.bls_g2_mul(a, kb)
        assert rc == rc2 == 0 and got == want, hex(k)


def test_gpu_g2_msm_parity_small(gpu, oracle_mod):
    for n in (1, 2, 100, 1024):
        pts = oracle_mod.bls_g2_gen_points(0, n)
        scs = oracle_mod.bls_gen_fr(51, n)
        rc, got = gpu.bls_g2_msm(pts, scs, n)
        rc2, want = oracle_mod.bls_g2_msm(pts, scs, n)
        assert rc == rc2 == 0 and got == want, n


def test_gpu_g2_msm_plan_parity_4096(gpu, oracle_mod):
    n = 4096
    plan = gpu.BlsG2MsmPlan(n)
    plan.gen_points(0)
    pts = plan.download_points()
    scs = gpu.bls_gen_fr(52, n)
    plan.upload_scalars(scs)
    got = plan.run()
    plan.destroy()
    rc, want = oracle_mod.bls_g2_msm(pts, scs, n)
    assert rc == 0 and got == want


def test_gpu_g2_msm_plan_parity_2_17_c16(gpu, oracle_mod):
    """n > 2^16 switches the G2 plan to the c=16 window config — the only
    test that exercises the large-window G2 instantiation."""
    n = 1 << 17
    plan = gpu.BlsG2MsmPlan(n)
    plan.gen_points(0)
    pts = plan.download_points()
    scs = gpu.bls_gen_fr(55, n)
    plan.upload_scalars(scs)
    got = plan.run()
    plan.destroy()
    rc, want = oracle_mod.bls_g2_msm(pts, scs, n)
    assert rc == 0 and got == want


def test_gpu_g2_msm_identity_and_zero_scalars(gpu, oracle_mod):
    n = 256
    pts = bytearray(oracle_mod.bls_g2_gen_points(0, n))
    scs = bytearray(oracle_mod.bls_gen_fr(53, n))
    pts[192 * 7:192 * 8] = b"\x00" * 192   # identity point
    scs[32 * 3:32 * 4] = b"\x00" * 32      # zero scalar
    pts, scs = bytes(pts), bytes(scs)
    rc, got = gpu.bls_g2_msm(pts, scs, n)
    rc2, want = oracle_mod.bls_g2_msm(pts, scs, n)
    assert rc == rc2 == 0 and got == want


def test_gpu_g2_msm_rejects_invalid(gpu, oracle_mod):
    pts = oracle_mod.bls_g2_gen_points(0, 1)
    bad = bytearray(pts)
    bad[191] ^= 1  # off curve
    rc, _ = gpu.bls_g2_msm(bytes(bad), (5).to_bytes(32, "big"), 1)
    assert rc == gpu.EM_ERR_POINT
    P = int("1a0111ea397fe69a4b1ba7b6434bacd764774b84f38512bf6730d2a0f6b0f624"
            "1eabfffeb153ffffb9feffffffffaaab", 16)
    noncanon = P.to_bytes(48, "big") + b"\x00" * 144
    rc, _ = gpu.bls_g2_msm(noncanon, (5).to_bytes(32, "big"), 1)
    assert rc == gpu.EM_ERR_INPUT


def test_gpu_g2_async_pipeline_matches_sync(gpu, oracle_mod):
    n = 2048
    plan = gpu.BlsG2MsmPlan(n)
    plan.gen_points(0)
    plan.upload_scalars(gpu.bls_gen_fr(54, n))
    want = plan.run()
    for _ in range(3):
        plan.run_async()
    got = plan.sync()
    plan.destroy()
    assert got == want


def test_gpu_g2_upload_roundtrip(gpu, oracle_mod):
    n = 128
    pts = oracle_mod.bls_g2_gen_points(5, n)
    plan = gpu.BlsG2MsmPlan(n)
    plan.upload_points(pts)
    assert plan.download_points() == pts
    plan.destroy()
