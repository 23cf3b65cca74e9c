"""Multi-process CPU test of the N>1 exchange path (gloo, world_size=2):
the exact allgather_partials code bench.py uses under RCCL, with oracle
partials standing in for the GPU partials, checked against the unsharded
oracle MSM (SURVEY.md §8e shard identity)."""
import os

import pytest

torch = pytest.importorskip("torch")


def _worker(rank, world, port, results_q):
    import torch.distributed as tdist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    tdist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import oracle as orc
        from ethrex_amd.dist import allgather_partials, shard_range

        n_total = 512
        lo, hi = shard_range(n_total, world, rank)
        pts = orc.gen_points(lo, hi - lo)
        scs = orc.gen_fr(42 + rank, hi - lo)
        rc, partial = orc.g1_msm_jacobian(pts, scs, hi - lo)
        assert rc == 0
        allparts = allgather_partials(partial, tdist, device="cpu")
        rc, combined = orc.g1_combine_jacobian(allparts, world)
        assert rc == 0
        # the product path's HOST combine (the pipelined N>1 loop's
        # per-step combine, bench.py) must agree with the oracle's
        import ethrex_amd
        assert ethrex_amd.g1_combine_cpu(allparts, world) == combined
        if rank == 0:
            # unsharded reference: same points, concatenated per-rank scalars
            all_pts, all_scs = b"", b""
            for r in range(world):
                rlo, rhi = shard_range(n_total, world, r)
                all_pts += orc.gen_points(rlo, rhi - rlo)
                all_scs += orc.gen_fr(42 + r, rhi - rlo)
            rc, want = orc.g1_msm(all_pts, all_scs, n_total)
            assert rc == 0
            results_q.put(("ok", combined == want))
        else:
            results_q.put(("ok", True))
    except Exception as e:  # surface worker failures to the test
        results_q.put(("err", repr(e)))
        raise
    finally:
        tdist.destroy_process_group()


def test_sharded_exchange_matches_unsharded():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for tag, val in outs:
        assert tag == "ok", val
        assert val is True


def test_shard_range_covers():
    from ethrex_amd.dist import shard_range
    for n, w in [(16, 1), (17, 4), (1 << 20, 8), (10, 3)]:
        spans = [shard_range(n, w, r) for r in range(w)]
        assert spans[0][0] == 0 and spans[-1][1] == n
        for a, b in zip(spans, spans[1:]):
            assert a[1] == b[0]
