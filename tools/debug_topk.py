"""Diagnostic: run the headline query a few times and print the per-phase
breakdown from sdb_gpu_last_stats. Usage (GPU box): python tools/debug_topk.py [docs]"""
import ctypes as CT
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

docs = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000_000
sels = [0.10, 0.05, 0.02, 0.01]
t0 = time.time()
cache = os.environ.get("SDB_BLOB_CACHE")
if cache and os.path.exists(cache):
    blob = open(cache, "rb").read()
    print(f"loaded cached blob {len(blob)/1e6:.1f}MB")
else:
    blob = sa.build_synth_segment(43, 1, docs, sels)
    print(f"build {time.time()-t0:.1f}s blob {len(blob)/1e6:.1f}MB")
    if cache:
        open(cache, "wb").write(blob)
ctx = sa.GpuContext(0)
seg = ctx.load_segment(blob)
lib = sa.gpu()
lib.sdb_gpu_last_stats.restype = CT.c_int
for it in range(6):
    t0 = time.time()
    hits, total = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 1000)
    wall = (time.time() - t0) * 1000
    km = CT.c_double(); nc = CT.c_uint(); gt = CT.c_float()
    rb = CT.c_double(); sel = CT.c_double()
    lib.sdb_gpu_last_stats(ctx._ctx, CT.byref(km), CT.byref(nc),
                           CT.byref(gt), CT.byref(rb), CT.byref(sel))
    print(f"it{it}: wall={wall:8.2f}ms kernel={km.value:7.2f}ms "
          f"readback={rb.value:7.2f}ms select={sel.value:7.2f}ms "
          f"ncand={nc.value} gtau={gt.value:.4f} total={total}")
