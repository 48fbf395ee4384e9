#!/usr/bin/env python3
"""Row-materialization (next_batch / PgFetchNext) micro-bench: the emit
path VERDICT r01 flagged at 1 wave/SIMD before the dynamic-LDS change."""
import ctypes as C
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import ybgpu as y  # noqa: E402
from gpu_scan import GpuScan  # noqa: E402

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 10_000_000
schema = y.make_schema([y.KT_INT64],
                       [(10 + i, y.T_INT64, 1) for i in range(4)])
data, offsets, nb, total, ne = y.generate(schema, rows=rows, seed=42)
spec = y.ScanSpec()
spec.schema = schema
spec.kv_format = y.ENC_THREE_SHARED_PARTS
spec.read_time = y.read_time(1_700_000_000_000_000)
spec.num_preds = 1
spec.preds[0] = y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0)  # ~50% match
spec.emit_rows = 1
s = GpuScan(spec)
s.feed_blocks_host(data, offsets, nb, total)
b = s.batch_rows()  # warm (includes flags pre-pass + emit + D2H)
t0 = time.time()
reps = 5
for _ in range(reps):
    raw = y.RowBatch()
    rc = s._lib.yb_gpu_scan_next_batch(s._h, C.byref(raw))
    assert rc == 0
dt = (time.time() - t0) / reps
print(f"emit: {rows} rows scanned, {raw.n_rows} matched; "
      f"{dt*1e3:.2f} ms/batch incl. D2H = "
      f"{rows/dt/1e9:.2f} Grows/s scanned, "
      f"{raw.n_rows/dt/1e6:.1f} Mrows/s materialized")
s.close()
