"""GPU parity for the BLS12-381 G1 MSM (the blob-KZG
commitment MSM / EIP-2537 G1 MSM): HIP path vs the CPU oracle, bit-exact,
including the blob-sized 4096-point MSM the sequencer computes per blob
(crates/common/crypto/kzg.rs:208-230)."""
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    import ethrex_amd
    if ethrex_amd.device_count() < 1:
        pytest.skip("no GPU")
    ethrex_amd.set_device(0)
    return ethrex_amd


def _b(h):
    return bytes.fromhex(h)


def test_gpu_bls_g1_add_golden(gpu, golden):
    for v in golden["bls_g1_add"]:
        rc, out = gpu.bls_g1_add(_b(v["a"]), _b(v["b"]))
        if v.get("error") == "point":
            assert rc == gpu.EM_ERR_POINT, v["name"]
        elif v.get("error") == "input":
            assert rc == gpu.EM_ERR_INPUT, v["name"]
        else:
            assert rc == 0 and out == _b(v["out"]), v["name"]


def test_gpu_bls_g1_mul_golden(gpu, golden):
    for v in golden["bls_g1_mul"]:
        rc, out = gpu.bls_g1_mul(_b(v["point"]), _b(v["scalar"]))
        assert rc == 0 and out == _b(v["out"]), v["name"]


def test_gpu_bls_msm_golden(gpu, golden):
    for v in golden["bls_msm"]:
        rc, out = gpu.bls_g1_msm(_b(v["points"]), _b(v["scalars"]), v["n"])
        assert rc == 0 and out == _b(v["out"]), f"bls msm n={v['n']}"


def test_gpu_bls_msm_rejects_offsubgroup(gpu, golden):
    off = _b(golden["bls_offsubgroup_point"])
    rc, _ = gpu.bls_g1_msm(off, (1).to_bytes(32, "big"), 1)
    assert rc == gpu.EM_ERR_POINT
    rc, _ = gpu.bls_g1_add(off, b"\x00" * 96)
    assert rc == 0  # add accepts on-curve non-subgroup points


def test_gpu_bls_msm_blob_4096(gpu, oracle_mod):
    """THE blob-KZG commitment shape: 4096-point MSM with canonical Fr
    scalars (blob field elements)."""
    n = 4096
    plan = gpu.BlsMsmPlan(n)
    plan.gen_points(0)
    pts = plan.download_points()
    assert pts == oracle_mod.bls_gen_points(0, n), "device gen_points parity"
    scs = gpu.bls_gen_fr(42, n)
    assert scs == oracle_mod.bls_gen_fr(42, n), "scalar generator parity"
    plan.upload_scalars(scs)
    got = plan.run()
    t = plan.last_times()
    plan.destroy()
    rc, want = oracle_mod.bls_g1_msm(pts, scs, n)
    assert rc == 0 and got == want
    assert t["total_ms"] > 0


def test_gpu_bls_fixed_base_blob(gpu, oracle_mod):
    """fixed-base (precomputed) blob commitment path: bit-exact vs the
    oracle and vs the non-precomputed path across multiple scalar sets."""
    n = 4096
    plan = gpu.BlsMsmPlan(n)
    plan.gen_points(0)
    pts = plan.download_points()
    plan.precompute()
    for seed in (42, 99):
        scs = gpu.bls_gen_fr(seed, n)
        plan.upload_scalars(scs)
        got = plan.run()
        rc, want = oracle_mod.bls_g1_msm(pts, scs, n)
        assert rc == 0 and got == want, seed
    plan.destroy()


def test_gpu_bls_msm_parity_2_14(gpu, oracle_mod):
    n = 1 << 14
    plan = gpu.BlsMsmPlan(n)
    plan.gen_points(0)
    pts = plan.download_points()
    scs = gpu.bls_gen_fr(7, n)
    plan.upload_scalars(scs)
    got = plan.run()
    plan.destroy()
    rc, want = oracle_mod.bls_g1_msm(pts, scs, n)
    assert rc == 0 and got == want


def test_gpu_bls_shard_combine(gpu, oracle_mod):
    n, shards = 2048, 4
    sh = n // shards
    parts = b""
    all_pts, all_scs = b"", b""
    for s in range(shards):
        plan = gpu.BlsMsmPlan(sh)
        plan.gen_points(s * sh)
        scs = gpu.bls_gen_fr(42 + s, sh)
        plan.upload_scalars(scs)
        parts += plan.run_partial()
        all_pts += plan.download_points()
        all_scs += scs
        plan.destroy()
    rc, got = gpu.bls_g1_combine(parts, shards)
    assert rc == 0
    rc, want = oracle_mod.bls_g1_msm(all_pts, all_scs, n)
    assert rc == 0 and got == want
    rc, want2 = oracle_mod.bls_g1_combine_jacobian(parts, shards)
    assert rc == 0 and got == want2


def test_gpu_bls_upload_points_roundtrip(gpu, oracle_mod):
    """uploaded host points parse+download bit-exactly (canonical IO)."""
    n = 256
    pts = oracle_mod.bls_gen_points(100, n)
    plan = gpu.BlsMsmPlan(n)
    plan.upload_points(pts)
    assert plan.download_points() == pts
    plan.destroy()
