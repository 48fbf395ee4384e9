"""Multi-process CPU test of the distributed path (world_size 2, gloo):
tablet->rank sharding + the final cross-tablet aggregate all-reduce — the
exact partitioning and collective bench.py uses at N>1 (tablets
are disjoint hash-range partitions; one tiny all-reduce merges partial
aggregates). Compute here is the CPU oracle (test infrastructure) since this
container has no GPU; the GPU parity suite covers the same per-tablet scan on
the device."""
import os
import sys

import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, port, out_q):
    sys.path.insert(0, ROOT)
    import torch
    import torch.distributed as dist
    import ybgpu as y

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    n_tablets = 8
    rows_per_tablet = 2000
    my_tablets = [t for t in range(n_tablets) if t % world == rank]

    total_sum = 0
    total_cnt = 0
    for t in my_tablets:
        data, offsets, nb, total, ne = y.generate(
            schema, rows=rows_per_tablet, seed=42 + t, nthreads=1)
        osc = y.orcl_schema_from(schema)
        spec = y.OrclScanSpec()
        spec.read_time = y.orcl_read_time(1_700_000_000_000_000)
        spec.num_preds = 1
        spec.preds[0] = y.OrclPred(0, 0, y.PRED_GT, 1 << 39, None, 0)
        spec.num_aggs = 2
        spec.aggs[0] = y.OrclAgg(y.AGG_SUM_INT64, 3)
        spec.aggs[1] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
        res, _ = y.orcl_scan(data, offsets, nb, osc, spec)
        total_sum += res.aggs[0].value_i64
        total_cnt += res.aggs[1].value_i64

    buf = torch.tensor([total_sum, total_cnt], dtype=torch.int64)
    dist.all_reduce(buf)  # the one cross-tablet collective
    if rank == 0:
        out_q.put((int(buf[0].item()), int(buf[1].item())))
    dist.destroy_process_group()


def test_tablet_sharding_allreduce():
    import ybgpu as y

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=300)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # reference: scan all 8 tablets in one process
    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    tot_sum = 0
    tot_cnt = 0
    for t in range(8):
        data, offsets, nb, total, ne = y.generate(
            schema, rows=2000, seed=42 + t, nthreads=1)
        osc = y.orcl_schema_from(schema)
        spec = y.OrclScanSpec()
        spec.read_time = y.orcl_read_time(1_700_000_000_000_000)
        spec.num_preds = 1
        spec.preds[0] = y.OrclPred(0, 0, y.PRED_GT, 1 << 39, None, 0)
        spec.num_aggs = 2
        spec.aggs[0] = y.OrclAgg(y.AGG_SUM_INT64, 3)
        spec.aggs[1] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
        res, _ = y.orcl_scan(data, offsets, nb, osc, spec)
        tot_sum += res.aggs[0].value_i64
        tot_cnt += res.aggs[1].value_i64

    assert got == (tot_sum, tot_cnt)


def _group_worker(rank, world, port, out_q):
    sys.path.insert(0, ROOT)
    import torch.distributed as dist
    import ybgpu as y

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    schema = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    my_tablets = [t for t in range(8) if t % world == rank]
    partial = {}
    for t in my_tablets:
        data, offsets, nb, total, ne = y.generate(
            schema, rows=1500, seed=42 + t, nthreads=1, group_mod=64)
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(1_700_000_000_000_000)
        spec.group_col = 1
        spec.num_aggs = 2
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 1)
        for key, row in y.sim_group(spec, data, offsets, nb).items():
            # row = [COUNT, SUM] (None when no contribution)
            c0 = row[0] or 0
            s1 = row[1] or 0
            if key in partial:
                partial[key] = (partial[key][0] + c0, partial[key][1] + s1)
            else:
                partial[key] = (c0, s1)

    # the cross-tablet GROUP-TABLE merge: all_gather the partial tables,
    # fold in rank order (SURVEY §8e — config 5's collective)
    gathered = [None] * world
    dist.all_gather_object(gathered, partial)
    if rank == 0:
        merged = {}
        for part in gathered:
            for key, (c0, s1) in part.items():
                mc, ms = merged.get(key, (0, 0))
                merged[key] = (mc + c0, ms + s1)
        out_q.put(merged)
    dist.destroy_process_group()


def test_group_table_merge_across_ranks():
    """Config-5 semantics at N>1: per-rank GROUP BY partial tables merged
    across ranks equal the oracle's grouped result over all 8 tablets."""
    import ybgpu as y

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_group_worker, args=(r, 2, 29523, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=300)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    schema = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    want = {}
    for t in range(8):
        data, offsets, nb, total, ne = y.generate(
            schema, rows=1500, seed=42 + t, nthreads=1, group_mod=64)
        osc = y.orcl_schema_from(schema)
        ospec = y.OrclScanSpec()
        ospec.read_time = y.orcl_read_time(1_700_000_000_000_000)
        ospec.num_aggs = 2
        ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
        ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 1)
        for key, row in y.orcl_group(
                data, offsets, nb, osc, ospec, group_col=0,
                num_aggs=2, aggs=ospec.aggs).items():
            c0 = row[0] or 0
            s1 = row[1] or 0
            wc, ws = want.get(key, (0, 0))
            want[key] = (wc + c0, ws + s1)
    assert got == want
