"""World-size-2 gloo tests (CPU, run in CI containers): the distributed
identities the N>1 GPU path relies on, exercised through the real
torch.distributed stack that bench.py uses for rendezvous.

 - cdbhash sharding is a partition (every row owned by exactly one
   segment; shard partials sum to the global answer — the 2-stage agg
   identity, cdbgroup.c:1245)
 - the count-matrix transpose used for alltoallv receive layouts
 - top-k merge of per-rank top-k lists equals global top-k
"""
import multiprocessing as mp
import os

import pytest


def _worker(rank, world, q, port=29511):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle"))
    import pyoracle
    import torch

    try:
        cutoff = pyoracle.pgdate(1998, 8, 15)
        # each rank computes ITS segment partial of Q1 on SF1
        mine = pyoracle.q1_synth_segment(42, 1, world, rank, cutoff)
        flat = []
        for g in mine:
            flat += [g["count"], g["sum_qty_c"], g["sum_base_c"],
                     g["sum_dcol_c"],
                     g["sum_disc4"] & ((1 << 63) - 1),
                     g["sum_charge6"] & ((1 << 63) - 1)]
        t = torch.tensor(flat, dtype=torch.int64)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        full = pyoracle.q1_synth(42, 1, cutoff)
        expect = []
        for g in full:
            expect += [g["count"], g["sum_qty_c"], g["sum_base_c"],
                       g["sum_dcol_c"],
                       g["sum_disc4"] & ((1 << 63) - 1),
                       g["sum_charge6"] & ((1 << 63) - 1)]
        assert t.tolist() == expect, "segment partials != global"

        # count-matrix transpose (the alltoallv receive layout rule in
        # exec_q3: rcnts[s] = allcnt[s * nseg + me])
        send_cnts = [10 * rank + d for d in range(world)]
        allcnt = [torch.zeros(world, dtype=torch.int64)
                  for _ in range(world)]
        dist.all_gather(allcnt, torch.tensor(send_cnts, dtype=torch.int64))
        rcnts = [int(allcnt[s][rank]) for s in range(world)]
        assert rcnts == [10 * s + rank for s in range(world)]

        # per-rank top-k lists merge to the global top-k
        import random
        rng = random.Random(7)
        rows = [(rng.randrange(10**9), rng.randrange(1000), i)
                for i in range(1000)]
        owned = [r for r in rows
                 if pyoracle.lib().gg_oracle_segment_int8(r[2] + 1, world)
                 == rank]
        localtop = sorted(owned, key=lambda r: (-r[0], r[1], r[2]))[:10]
        gathered = [None] * world
        dist.all_gather_object(gathered, localtop)
        merged = sorted(sum(gathered, []),
                        key=lambda r: (-r[0], r[1], r[2]))[:10]
        expect_top = sorted(rows, key=lambda r: (-r[0], r[1], r[2]))[:10]
        assert merged == expect_top
        q.put((rank, "ok"))
    except Exception as exc:  # noqa: BLE001
        q.put((rank, f"FAIL: {exc}"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_gloo_world2_distributed_identities():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=150) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", (rank, status)


@pytest.mark.timeout(240)
def test_gloo_world4_distributed_identities():
    """Same identities at world size 4 — the driver's 4/8-GPU scale
    runs exercise the same sharding/transpose/merge rules."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 4, q, 29517))
             for r in range(4)]
    for p in procs:
        p.start()
    results = [q.get(timeout=200) for _ in range(4)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", (rank, status)
