"""Scan kernel probe: min step time on 1B rows."""
import ctypes as CT, os, sys, time
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

rows, ngroups = 1_000_000_000, 1024
rng = np.random.default_rng(44)
keys = rng.integers(0, ngroups, rows).astype(np.int64)
v1 = rng.integers(0, 1 << 20, rows).astype(np.int64)
v2 = rng.normal(0, 1, rows).astype(np.float32)
ctx = sa.GpuContext(0); lib = sa.gpu()
class ColView(CT.Structure):
    _fields_ = [("data", CT.c_void_p), ("rows", CT.c_uint64), ("type", CT.c_int)]
class PredSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int), ("ilo", CT.c_int64),
                ("ihi", CT.c_int64), ("flo", CT.c_float), ("fhi", CT.c_float)]
class AggSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int)]
class AggResult(CT.Structure):
    _fields_ = [("i64", CT.c_int64), ("f64", CT.c_double)]
cols = (ColView * 3)(ColView(keys.ctypes.data_as(CT.c_void_p).value, rows, 0),
                     ColView(v1.ctypes.data_as(CT.c_void_p).value, rows, 0),
                     ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
tab = CT.c_void_p(0)
assert lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(rows), CT.byref(tab)) == 0
preds = (PredSpec * 1)(PredSpec(1, 1, int((1 << 20) * 0.1), 0, 0, 0))
aggs = (AggSpec * 3)(AggSpec(0, 0), AggSpec(1, 1), AggSpec(2, 2))
out = (AggResult * (ngroups * 3))(); passed = CT.c_uint64(0)
def step():
    assert lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs, 3, out, CT.byref(passed)) == 0
for _ in range(3): step()
best = 1e9
for _ in range(8):
    t0 = time.time(); step(); best = min(best, time.time() - t0)
print(f"best step {best*1000:.2f}ms -> {rows/best/1e9:.1f}G rows/s, {rows*20/best/1e12:.2f} TB/s")
