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


class _FakeShardPlan:
    """Stands in for the GPU MsmPlan in the pipelined-loop control-flow
    test: run_partial_async enqueues this rank's oracle partial for the
    step's seeded scalars; wait_one delivers the OLDEST pending one
    (depth-2 backpressure shape like the C side)."""

    def __init__(self, rank, world, n_total, steps):
        import oracle as orc
        self.pending = []
        from ethrex_amd.dist import shard_range
        lo, hi = shard_range(n_total, world, rank)
        pts = orc.gen_points(lo, hi - lo)
        self.parts = []
        for s in range(steps):
            scs = orc.gen_fr(100 + s * 10 + rank, hi - lo)
            rc, part = orc.g1_msm_jacobian(pts, scs, hi - lo)
            assert rc == 0
            self.parts.append(part)
        self.k = 0

    def run_partial_async(self):
        assert len(self.pending) < 2, "depth-2 backpressure violated"
        self.pending.append(self.parts[self.k])
        self.k += 1

    def wait_one(self):
        return self.pending.pop(0)


def _worker_pipelined(rank, world, port, results_q):
    import torch.distributed as tdist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    tdist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import oracle as orc
        from ethrex_amd.dist import pipelined_shard_steps, shard_range

        n_total, steps = 384, 4
        plan = _FakeShardPlan(rank, world, n_total, steps)
        got = []
        last = pipelined_shard_steps(plan, steps, tdist, world,
                                     collect=got)
        assert last == got[-1] and len(got) == steps
        # every step's combined result == the unsharded oracle MSM over
        # that step's per-rank-seeded scalars
        ok = True
        for s in range(steps):
            all_pts, all_scs = b"", b""
            for r in range(world):
                rlo, rhi = shard_range(n_total, world, r)
                all_pts += orc.gen_points(rlo, rhi - rlo)
                all_scs += orc.gen_fr(100 + s * 10 + r, rhi - rlo)
            rc, want = orc.g1_msm(all_pts, all_scs, n_total)
            ok = ok and rc == 0 and got[s] == want
        results_q.put(("ok", ok))
    except Exception as e:
        results_q.put(("err", repr(e)))
        raise
    finally:
        tdist.destroy_process_group()


def test_pipelined_shard_loop_control_flow():
    """World-2 gloo test of bench.py's N>1 timed loop (the driver's SCALE
    harness path): per-step delivery order, depth-2 backpressure, gather
    + host combine — each step's result equals the unsharded oracle."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_pipelined, args=(r, 2, 29519, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for tag, val in outs:
        assert tag == "ok", val
        assert val is True
