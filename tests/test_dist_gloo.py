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
