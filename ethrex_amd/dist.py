"""Multi-GPU sharding helpers: point-index sharding with an RCCL (xGMI)
AllGather of the 96-byte Jacobian G1 partials + local combine (SURVEY.md
§8e: EC addition is not an RCCL ReduceOp, so the exchange is an AllGather
of the tiny partials; latency-bound, compute fully overlaps).

The exchange is torch.distributed so the identical code path runs under
"nccl" (=RCCL) on GPUs and "gloo" on CPU in the world_size-2 tests.
"""


def shard_range(n_total: int, world: int, rank: int):
    """Contiguous point-index shard for this rank."""
    shard = n_total // world
    extra = n_total % world
    lo = rank * shard + min(rank, extra)
    hi = lo + shard + (1 if rank < extra else 0)
    return lo, hi


def pipelined_shard_steps(plan, steps, tdist, world, group=None,
                          combine=None, collect=None, device="cpu"):
    """The N>1 timed loop (bench.py): enqueue step k, deliver step k-1's
    Jacobian partial (plan.wait_one leaves step k running on the GPU),
    AllGather it and combine on the host — the exchange of step k-1
    fully overlaps the GPU compute of step k.

    `plan` needs run_partial_async() and wait_one(); `combine(bytes,
    world) -> bytes` defaults to the host combine.  Returns the last
    step's combined result; appends every step's result to `collect`
    when given (tests)."""
    if combine is None:
        from .lib import g1_combine_cpu
        combine = g1_combine_cpu
    last = None
    plan.run_partial_async()
    for _ in range(1, steps):
        plan.run_partial_async()
        part = plan.wait_one()
        allp = allgather_partials(part, tdist, device=device, group=group)
        last = combine(allp, world)
        if collect is not None:
            collect.append(last)
    part = plan.wait_one()
    allp = allgather_partials(part, tdist, device=device, group=group)
    last = combine(allp, world)
    if collect is not None:
        collect.append(last)
    return last


def allgather_partials(partial96: bytes, tdist, device="cpu", group=None):
    """AllGather each rank's 96-B Jacobian partial; returns concatenated
    world*96 bytes in rank order.

    The payload is delivered to the HOST by msm_wait_one, so the N>1
    bench path gathers it over a gloo subgroup (device="cpu"): 96 B over
    loopback/shm costs ~0.1 ms and — unlike a cuda-tensor gather — adds
    no H2D/D2H hops and never touches the GPU streams mid-pipeline."""
    import torch
    assert len(partial96) == 96
    world = tdist.get_world_size(group) if group else tdist.get_world_size()
    t_in = torch.frombuffer(bytearray(partial96), dtype=torch.uint8).to(device)
    t_out = torch.empty(world * 96, dtype=torch.uint8, device=device)
    tdist.all_gather_into_tensor(t_out, t_in, group=group)
    return bytes(t_out.cpu().numpy().tobytes())
