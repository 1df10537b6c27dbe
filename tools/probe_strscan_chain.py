"""Chain-vs-single hybrid cost + string-keyed scan throughput (informational)."""
import os, sys, time
import numpy as np
sys.path.insert(0, "/root/repo")
import serenedb_amd as sa
import ctypes as CT

# hybrid single vs 2-pred chain, 100M docs
doc_count = 100_000_000
blob = sa.build_synth_segment(43, 1, doc_count, [0.10, 0.05, 0.02, 0.01])
ctx = sa.GpuContext(0)
seg = ctx.load_segment(blob)
rng = np.random.default_rng(45)
span = 1 << 31
col0 = rng.integers(0, span, doc_count + 1).astype(np.int64)
col1 = rng.integers(0, span, doc_count + 1).astype(np.int64)
ctx.attach_column(seg, col0, slot=0)
ctx.attach_column(seg, col1, slot=1)
flo, fhi = int(span * 0.4), int(span * 0.6) - 1

def best(f, reps=15):
    b = 1e9
    for _ in range(reps):
        t0 = time.time(); f(); b = min(b, time.time() - t0)
    return b * 1e3

t1 = best(lambda: ctx.execute_topk_hybrid([seg], [0,1,2,3], [1.0]*4, 1000,
                                          flo, fhi, 64))
t2 = best(lambda: ctx.execute_topk_hybrid_chain(
    [seg], [0,1,2,3], [1.0]*4, 1000,
    [(0, 3, flo, fhi), (1, 2, span // 2, 0)], 64))
print(f"hybrid single-pred: {t1:.3f} ms; 2-pred chain: {t2:.3f} ms "
      f"(+{100*(t2-t1)/t1:.1f}%)")

# string-keyed scan, 400M rows dense codes
lib = sa.gpu()
n = 400_000_000
vocab = [f"{c}{i:03d}" for c in "abcdefgh" for i in range(64)]
kcodes = rng.integers(0, len(vocab), n).astype(np.int64)
fcodes = rng.integers(0, len(vocab), n).astype(np.int64)
v2 = rng.normal(0, 1, n).astype(np.float32)
op, lo, hi = sa.str_pred_to_code(vocab, "prefix", "c")

class ColView(CT.Structure):
    _fields_ = [("data", CT.c_void_p), ("rows", CT.c_uint64), ("type", CT.c_int)]
class PredSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int), ("ilo", CT.c_int64),
                ("ihi", CT.c_int64), ("flo", CT.c_float), ("fhi", CT.c_float)]
class AggSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int)]
class AggResult(CT.Structure):
    _fields_ = [("i64", CT.c_int64), ("f64", CT.c_double)]
cols = (ColView * 3)(
    ColView(kcodes.ctypes.data_as(CT.c_void_p).value, n, 0),
    ColView(fcodes.ctypes.data_as(CT.c_void_p).value, n, 0),
    ColView(v2.ctypes.data_as(CT.c_void_p).value, n, 1))
tab = CT.c_void_p(0)
assert lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(n), CT.byref(tab)) == 0
preds = (PredSpec * 1)(PredSpec(1, op, lo, hi, 0, 0))
aggs = (AggSpec * 2)(AggSpec(0, 0), AggSpec(1, 1))
out = (AggResult * (len(vocab) * 2))()
passed = CT.c_uint64(0)
def scan():
    assert lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, len(vocab), preds, 1,
                                aggs, 2, out, CT.byref(passed)) == 0
b = best(scan, 10)
print(f"string-keyed scan (512 groups, prefix pred): {b:.2f} ms "
      f"-> {n/b/1e6:.1f}G rows/s")
