"""World-size-2 gloo tests (CPU) of bench.py's distributed harness logic:
rank seeding, barrier + max-over-ranks timing, and whole-job aggregation.
The compute path itself needs the GPU (driver's round-end scaling run);
this pins the N>1 plumbing that wraps it (independent proofs
per rank, no data-path collective)."""
import multiprocessing as mp
import os

import pytest


def _worker(rank, world, q):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT="29571",
        RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    import torch
    import torch.distributed as dist

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    # per-rank seed derivation (same formula as bench.py's proof workload)
    SEED = 0x5441494741
    step = 3
    inst = (SEED + rank).to_bytes(16, "little") + step.to_bytes(16, "little")
    assert len(inst) == 32
    # simulated per-rank elapsed: max-over-ranks must pick the slowest
    elapsed = 1.0 + rank * 0.5
    t = torch.tensor([elapsed], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    dist.barrier()
    q.put((rank, float(t.item()), inst.hex()))
    dist.destroy_process_group()


def test_gloo_rank_harness():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, tmax, inst = q.get(timeout=120)
        results[rank] = (tmax, inst)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # every rank agreed on the max elapsed (rank 1's 1.5)
    assert all(abs(v[0] - 1.5) < 1e-9 for v in results.values())
    # ranks derived distinct instance seeds
    assert results[0][1] != results[1][1]
