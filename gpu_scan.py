"""GpuScan — thin Python wrapper over the product C ABI
(include/yb_gpu_scan.h). This is the PRODUCT path: it requires an MI355X
(gfx950) and fails loudly when no HIP device or extension is present — there
is no CPU fallback."""
import ctypes as C

import ybgpu as y


class GpuScanError(RuntimeError):
    pass


def _lib():
    lib = y.product()
    f = lib.yb_gpu_scan_open
    f.restype = C.c_int
    f.argtypes = [C.POINTER(y.ScanSpec), C.POINTER(C.c_void_p)]
    lib.yb_gpu_scan_feed_blocks.restype = C.c_int
    lib.yb_gpu_scan_feed_blocks.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.c_int]
    lib.yb_gpu_scan_execute.restype = C.c_int
    lib.yb_gpu_scan_execute.argtypes = [C.c_void_p]
    lib.yb_gpu_scan_wait.restype = C.c_int
    lib.yb_gpu_scan_wait.argtypes = [C.c_void_p]
    lib.yb_gpu_scan_aggregate.restype = C.c_int
    lib.yb_gpu_scan_aggregate.argtypes = [C.c_void_p,
                                          C.POINTER(y.ScanResult)]
    lib.yb_gpu_scan_next_batch.restype = C.c_int
    lib.yb_gpu_scan_next_batch.argtypes = [C.c_void_p,
                                           C.POINTER(y.RowBatch)]
    lib.yb_host_iter_open.restype = C.c_void_p
    lib.yb_host_iter_open.argtypes = [C.POINTER(y.ScanSpec),
                                      C.POINTER(C.c_uint8),
                                      C.POINTER(C.c_uint64), C.c_uint64]
    lib.yb_host_iter_next.restype = C.c_int
    lib.yb_host_iter_next.argtypes = [C.c_void_p, C.POINTER(C.c_uint64),
                                      C.POINTER(C.c_uint64),
                                      C.POINTER(C.c_uint32),
                                      C.POINTER(C.POINTER(C.c_uint8))]
    lib.yb_host_iter_close.restype = None
    lib.yb_host_iter_close.argtypes = [C.c_void_p]
    lib.yb_gpu_scan_group_aggregate.restype = C.c_int
    lib.yb_gpu_scan_group_aggregate.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint64), C.POINTER(C.c_int64),
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint8), C.c_uint64, C.c_uint64,
        C.POINTER(C.c_uint64)]
    lib.yb_gpu_scan_restart_data.restype = C.c_int
    lib.yb_gpu_scan_restart_data.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.POINTER(C.c_uint32)]
    lib.yb_gpu_scan_kernel_ms.restype = C.c_int
    lib.yb_gpu_scan_kernel_ms.argtypes = [C.c_void_p, C.POINTER(C.c_double),
                                          C.POINTER(C.c_double)]
    lib.yb_gpu_scan_close.restype = C.c_int
    lib.yb_gpu_scan_close.argtypes = [C.c_void_p]
    lib.yb_gpu_last_error.restype = C.c_char_p
    lib.yb_gpu_available.restype = C.c_int
    return lib


def gpu_available():
    return bool(_lib().yb_gpu_available())


class GpuScan:
    def __init__(self, spec):
        self._lib = _lib()
        self._h = C.c_void_p()
        self._spec = spec  # keep alive (aux pointers)
        rc = self._lib.yb_gpu_scan_open(C.byref(spec), C.byref(self._h))
        if rc:
            raise GpuScanError(
                f"yb_gpu_scan_open rc={rc}: "
                f"{self._lib.yb_gpu_last_error().decode()}")

    def _check(self, rc, what):
        if rc:
            raise GpuScanError(
                f"{what} rc={rc}: {self._lib.yb_gpu_last_error().decode()}")

    def feed_blocks_host(self, data, offsets, n_blocks, total_bytes):
        del total_bytes
        self._check(
            self._lib.yb_gpu_scan_feed_blocks(self._h, data, offsets,
                                              n_blocks, 0),
            "feed_blocks")

    def feed_blocks_intents(self, data, offsets, n_blocks, intents_blob,
                            blob_len, txns, n_txns):
        """Feed regular blocks + an intent stream (resolve + merge + feed;
        yb_gpu_scan_feed_blocks_intents)."""
        f = self._lib.yb_gpu_scan_feed_blocks_intents
        f.restype = C.c_int
        f.argtypes = [C.c_void_p, C.POINTER(C.c_uint8),
                      C.POINTER(C.c_uint64), C.c_uint64,
                      C.POINTER(C.c_uint8), C.c_uint64,
                      C.POINTER(y.TxnStatus), C.c_uint32]
        self._check(f(self._h, data, offsets, n_blocks, intents_blob,
                      blob_len, txns, n_txns), "feed_blocks_intents")

    def feed_blocks_bloom(self, data, offsets, n_blocks, filt):
        """feed_blocks + bloom consultation: a point scan whose pinned key
        prefix the filter proves absent feeds nothing and reports empty
        (yb_gpu_scan_feed_blocks_bloom)."""
        f = self._lib.yb_gpu_scan_feed_blocks_bloom
        f.restype = C.c_int
        f.argtypes = [C.c_void_p, C.POINTER(C.c_uint8),
                      C.POINTER(C.c_uint64), C.c_uint64, C.c_int,
                      C.POINTER(C.c_uint8), C.c_uint64]
        fb = (C.c_uint8 * len(filt)).from_buffer_copy(filt) if filt else None
        self._check(f(self._h, data, offsets, n_blocks, 0, fb,
                      len(filt) if filt else 0), "feed_blocks_bloom")

    def feed_sst(self, file_ptr, size, verify=True):
        """Feed a complete BlockBasedTable SST file (footer + index block
        parsed host-side; block checksums verified when verify)."""
        import ctypes as C
        f = self._lib.yb_gpu_scan_feed_sst
        f.restype = C.c_int
        f.argtypes = [C.c_void_p, C.POINTER(C.c_uint8), C.c_uint64, C.c_int]
        self._check(f(self._h, file_ptr, size, 1 if verify else 0),
                    "feed_sst")

    def execute(self):
        self._check(self._lib.yb_gpu_scan_execute(self._h), "execute")

    def wait(self):
        self._check(self._lib.yb_gpu_scan_wait(self._h), "wait")

    def aggregates(self):
        res = y.ScanResult()
        self._check(self._lib.yb_gpu_scan_aggregate(self._h, C.byref(res)),
                    "aggregate")
        return res

    def batch_rows(self):
        """Materialize matching rows (next_batch) and decode into python
        tuples sorted by sort_key."""
        b = y.RowBatch()
        self._check(self._lib.yb_gpu_scan_next_batch(self._h, C.byref(b)),
                    "next_batch")
        return y.decode_batch_rows(self._spec.schema, b.n_rows, b.sort_key,
                                   b.key_datums, b.datums, b.null_masks,
                                   b.varlen)

    def group_aggregate_raw(self, cap=1 << 20, key_bytes_cap=1 << 24):
        """GROUP BY partial aggregates: raw ctypes arrays
        (keys, vals, cnts, key_bytes, n_groups). Output buffers are
        allocated once and reused (zeroed ctypes allocation of the full
        capacity measured ~30 ms/step); only the first n_groups entries
        are written by the call."""
        gb = getattr(self, "_group_bufs", None)
        if gb is None or gb[0] != cap or gb[1] != key_bytes_cap:
            gb = (cap, key_bytes_cap,
                  (C.c_uint64 * cap)(),
                  (C.c_int64 * (cap * y.MAX_AGGS))(),
                  (C.c_uint64 * (cap * y.MAX_AGGS))(),
                  (C.c_uint8 * key_bytes_cap)())
            self._group_bufs = gb
        keys, vals, cnts, kb = gb[2], gb[3], gb[4], gb[5]
        n = C.c_uint64()
        self._check(
            self._lib.yb_gpu_scan_group_aggregate(
                self._h, keys, vals, cnts, kb, key_bytes_cap, cap,
                C.byref(n)), "group_aggregate")
        return keys, vals, cnts, kb, n.value

    def group_aggregate(self, cap=1 << 20, key_bytes_cap=1 << 24):
        """GROUP BY partial aggregates (spec.group_col must be set)."""
        keys, vals, cnts, kb, n = self.group_aggregate_raw(cap, key_bytes_cap)
        return y._decode_groups(self._spec.schema, self._spec.group_col - 1,
                                n, keys, vals, cnts, kb,
                                self._spec.num_aggs, self._spec.aggs)

    def restart_data(self):
        """Read-restart data of the last execute/group_aggregate
        (GetReadRestartData analog): encoded DocHybridTime bytes, b'' when
        no restart is needed."""
        ht = (C.c_uint8 * y.MAX_HT)()
        ln = C.c_uint32()
        self._check(
            self._lib.yb_gpu_scan_restart_data(self._h, ht, C.byref(ln)),
            "restart_data")
        return bytes(ht[:ln.value])

    def kernel_ms(self):
        total = C.c_double()
        decode = C.c_double()
        self._check(
            self._lib.yb_gpu_scan_kernel_ms(self._h, C.byref(total),
                                            C.byref(decode)), "kernel_ms")
        return total.value, decode.value

    def close(self):
        if self._h:
            self._lib.yb_gpu_scan_close(self._h)
            self._h = None
