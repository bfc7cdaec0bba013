"""Multi-rank engine worker (one process per rank, shared GPU).

Spawned by tests/test_gpu_multiproc.py (and tools/gpu_mp2.sh) to run
the REAL engine exchange path at world>1 on a single MI355X.  RCCL
refuses two communicator ranks on the same device, so the ranks
bootstrap over the engine's shm transport (comm.cpp, GG_COMM_SHM);
every Motion leg (alltoallv redistribute, allgather combine) still
executes the engine's own partition/scatter/receive-layout logic —
the code an 8-GPU RCCL run takes, minus only the ncclSend/Recv calls.

Usage: python tests/mp_engine_worker.py RANK WORLD IDFILE OUTFILE SF
"""
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    rank, world = int(sys.argv[1]), int(sys.argv[2])
    idfile, outfile, sf = sys.argv[3], sys.argv[4], int(sys.argv[5])

    from greengage_amd import Engine, PGDate
    from greengage_amd.engine import PIPE_Q1, PIPE_Q3, PIPE_Q5

    eng = Engine(device=0, n_segments=world, segment_id=rank)
    if rank == 0:
        cid = eng.comm_id()
        assert cid.startswith(b"GGSHM:"), "worker requires GG_COMM_SHM=1"
        with open(idfile + ".tmp", "wb") as f:
            f.write(cid)
        os.rename(idfile + ".tmp", idfile)
    else:
        for _ in range(600):
            if os.path.exists(idfile):
                break
            time.sleep(0.1)
        with open(idfile, "rb") as f:
            cid = f.read()
    eng.comm_init(cid)

    li = eng.register_synth("lineitem", seed=42, sf=sf)
    od = eng.register_synth("orders", seed=42, sf=sf)
    cu = eng.register_synth("customer", seed=42, sf=sf)
    su = eng.register_synth("supplier", seed=42, sf=sf)
    na = eng.register_synth("nation", seed=42, sf=sf)

    out = {"rank": rank, "world": world,
           "li_rows": eng.table_nrows(li),
           "od_rows": eng.table_nrows(od)}

    p1 = eng.compile(PIPE_Q1, lineitem=li, cutoff_date=PGDate("1998-08-15"))
    out["q1"] = eng.execute_q1(p1)

    p3 = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                     cutoff_date=PGDate("1995-03-15"), mktsegment=2,
                     limit_k=10)
    rows, hdr = eng.execute_q3(p3)
    # run twice: scratch reuse across executes must stay bit-exact
    rows2, hdr2 = eng.execute_q3(p3)
    assert rows == rows2 and hdr == hdr2, "Q3 not deterministic"
    out["q3_rows"], out["q3_hdr"] = rows, hdr

    p5 = eng.compile(PIPE_Q5, lineitem=li, orders=od, customer=cu,
                     supplier=su, nation=na,
                     cutoff_date=PGDate("1997-01-01"),
                     cutoff_hi=PGDate("1998-01-01"), regionkey=1)
    out["q5"] = eng.execute_q5(p5)

    # generic-descriptor plans at world>1: exercises exec_plan's
    # partial-group allgather combine (2-stage agg)
    from greengage_amd import pgdate
    from greengage_amd.engine import NEG_INF
    lo, hi = pgdate(1994, 1, 1), pgdate(1995, 1, 1)
    pq6 = eng.compile_plan(
        li, preds=[("shipdate", lo, hi), ("disc", 5, 8),
                   ("qty", NEG_INF, 2400)],
        aggs=[("sum", [("price", "id"), ("disc", "id")]), "count"])
    out["plan_q6"] = eng.execute_plan(pq6, max_groups=8)
    pg = eng.compile_plan(
        li, preds=[("shipdate", NEG_INF, PGDate("1998-08-15") + 1)],
        group_cols=["rflag", "lstatus"],
        aggs=["count", ("sum", [("qty", "id")])])
    out["plan_grouped"] = eng.execute_plan(pg, max_groups=64)
    # repeat execute: the second run takes the baked-codes kernel
    # (codes observed from the MERGED cross-segment group set) and
    # must still combine to the identical result
    out["plan_grouped_repeat"] = eng.execute_plan(pg, max_groups=64)

    eng.shutdown()
    with open(outfile + ".tmp", "w") as f:
        json.dump(out, f)
    os.rename(outfile + ".tmp", outfile)


if __name__ == "__main__":
    main()
