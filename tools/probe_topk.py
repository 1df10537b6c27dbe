"""Scaling probe: kernel time vs #terms (fixed corpus)."""
import ctypes as CT, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

cache = os.environ.get("SDB_BLOB_CACHE")
if cache and os.path.exists(cache):
    blob = open(cache, "rb").read()
else:
    blob = sa.build_synth_segment(43, 1, 100_000_000, [0.10, 0.05, 0.02, 0.01])
    if cache: open(cache, "wb").write(blob)
ctx = sa.GpuContext(0); seg = ctx.load_segment(blob)
lib = sa.gpu(); lib.sdb_gpu_last_stats.restype = CT.c_int
for terms in ([0], [3], [0,1], [0,1,2], [0,1,2,3]):
    for _ in range(3):
        ctx.execute_topk([seg], terms, [1.0]*len(terms), 1000)
    km = CT.c_double()
    best = 1e9
    for _ in range(5):
        ctx.execute_topk([seg], terms, [1.0]*len(terms), 1000)
        lib.sdb_gpu_last_stats(ctx._ctx, CT.byref(km), None, None, None, None)
        best = min(best, km.value)
    print(f"terms={terms} kernel={best:.3f}ms")
