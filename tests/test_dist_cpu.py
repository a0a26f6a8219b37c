"""Multi-process (gloo, world_size=2) test of the distributed lnL protocol:
contiguous site sharding + one all-reduce of the per-rank partial lnL — the
xGMI/RCCL path's logic, runnable on CPU (the GPU path swaps gloo->nccl and
the oracle executor->HIP engine, nothing else)."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, width, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import examl_amd as ea
        from tests.helpers import make_synthetic, oracle_full_lnl

        ntips = 12
        tips, wgt = make_synthetic(ntips, width, seed=2024)
        model = ea.DnaGtrModel([0.3, 0.2, 0.25, 0.25],
                               [1.0, 2.5, 0.7, 1.1, 3.0, 1.0], alpha=0.9)
        tree = ea.PhyloTree.random(ntips, seed=4, rng_z=True)
        entries, root = tree.full_traversal()

        # contiguous {offset, width} shard, the Kassian assignment shape
        # (partitionAssignment.c:398) for one partition over `world` ranks
        per = (width + world - 1) // world
        lo, hi = rank * per, min((rank + 1) * per, width)
        lnl = oracle_full_lnl(entries, root, tree, model,
                              np.ascontiguousarray(tips[:, lo:hi]),
                              wgt[lo:hi])
        t = torch.tensor([lnl], dtype=torch.float64)
        dist.all_reduce(t)  # replaces evaluateGenericSpecial.c:969
        if rank == 0:
            full = oracle_full_lnl(entries, root, tree, model, tips, wgt)
            result_q.put((t.item(), full))
    finally:
        dist.destroy_process_group()


def test_sharded_lnl_allreduce_matches_full():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, 2, 384, q)) for r in range(2)]
    for p in ps:
        p.start()
    reduced, full = q.get()
    for p in ps:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert np.isclose(reduced, full, rtol=1e-12)
