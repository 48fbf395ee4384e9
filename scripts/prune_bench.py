import ctypes as C, struct, sys, time
sys.path.insert(0, '/root/repo')
import ybgpu as y
from gpu_scan import GpuScan
schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1), (11, y.T_INT64, 1)], has_hash=False)
b = y.Builder(schema)
seq = 1 << 50
N = 2_000_000
for r in range(N):
    seq += 1
    b.add_packed_row(1000, [(y.T_INT64, r), (y.T_INT64, r * 3)], key_datums=(r,), seq=seq)
built = b.finish()
opts = [5, 777777, 1999998]
blob = b"".join(struct.pack("<q", v) for v in opts)
buf = C.create_string_buffer(blob, len(blob))
spec = y.ScanSpec(); spec.schema = schema
spec.kv_format = y.ENC_THREE_SHARED_PARTS
spec.read_time = y.read_time(5000)
spec.num_preds = 1
spec.preds[0] = y.Pred(1, 0, y.PRED_IN, 0, C.cast(buf, C.POINTER(C.c_uint8)), len(blob))
spec.num_aggs = 2
spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0); spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 1)
s = GpuScan(spec)
s.feed_blocks_host(built[0], built[1], built[2], built[3])
for _ in range(3): s.execute(); s.wait()
t0 = time.time()
for _ in range(20): s.execute(); s.wait()
dt = (time.time() - t0) / 20
res = s.aggregates(); tot, dec = s.kernel_ms()
print(f"mode={'pruned' if not __import__('os').environ.get('YBG_NOPRUNE') else 'brute'} "
      f"matched={res.rows_matched} sum={res.aggs[1].value_i64} entries={res.entries_seen} "
      f"step_ms={dt*1e3:.3f} kernel_ms={tot:.3f}")
s.close()
