"""C-ABI surface: the product library loads and exports every function
`include/yb_gpu_scan.h` declares (no compute calls — runs without a GPU),
and the no-GPU error path fails loudly (the product path has no CPU
fallback)."""
import ctypes as C
import os
import re

import ybgpu as y

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(ROOT, "include", "yb_gpu_scan.h")


def _declared_functions():
    src = open(HEADER).read()
    # strip comments
    src = re.sub(r"/\*.*?\*/", " ", src, flags=re.S)
    names = re.findall(
        r"^[A-Za-z_][\w \t\*]*?\b(yb_gpu_\w+|ybg_\w+)\s*\(", src, re.M)
    return sorted(set(names))


def test_every_header_symbol_exported():
    lib = y.product()
    decls = _declared_functions()
    assert len(decls) >= 20, decls  # the surface is substantial
    missing = [n for n in decls if not hasattr(lib, n)]
    assert not missing, f"header-declared but not exported: {missing}"


def test_host_iterator_symbols_exported():
    lib = y.product()
    for n in ("yb_host_iter_open", "yb_host_iter_next",
              "yb_host_iter_paging_state", "yb_host_iter_close"):
        assert hasattr(lib, n), n


def test_no_gpu_open_fails_loudly():
    """On a GPU-less host, yb_gpu_scan_open must return an error (there is
    no CPU fallback on the product path)."""
    lib = y.product()
    avail = lib.yb_gpu_available()
    if avail:
        return  # running on a GPU box: covered by the gpu suite
    spec = y.ScanSpec()
    spec.schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    spec.read_time = y.read_time(1_000_000)
    h = C.c_void_p()
    rc = lib.yb_gpu_scan_open(C.byref(spec), C.byref(h))
    assert rc != 0
    err = C.cast(lib.yb_gpu_last_error, C.CFUNCTYPE(C.c_char_p))()
    assert b"no HIP device" in err or b"fallback" in err, err
