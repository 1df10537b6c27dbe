#!/usr/bin/env python3
"""Hash-aggregate perf probe (dense kernel vs hash kernel, dense and
sparse keys)."""
import sys, os, time
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

rows = 400_000_000
rng = np.random.default_rng(44)
keys = rng.integers(0, 1024, rows).astype(np.int64)
v1 = rng.integers(0, 1 << 20, rows).astype(np.int64)
v2 = rng.normal(0, 1, rows).astype(np.float32)
ctx = sa.GpuContext(0)
tab = ctx.load_table([keys, v1, v2])
c = int((1 << 20) * 0.1)
arms = [
    ("dense", lambda: ctx.scan_agg(tab, 0, 1024, [(1, 1, c, 0)],
                                   [(0, 0), (1, 1), (2, 2)])),
    ("hash-dense-keys", lambda: ctx.scan_agg_hash(
        tab, 0, 2048, [(1, 1, c, 0)], [(0, 0), (1, 1), (2, 2)])),
]
for name, fn in arms:
    fn()
    t0 = time.time(); n = 4
    for _ in range(n): fn()
    dt = (time.time() - t0) / n
    print(f"{name}: {dt*1000:.2f} ms/pass = {rows/dt/1e9:.1f}G rows/s",
          flush=True)
skeys = rng.integers(-(1 << 60), 1 << 60, 100_000).astype(np.int64)[
    rng.integers(0, 100_000, rows)]
tab2 = ctx.load_table([skeys, v1, v2])
def hs():
    return ctx.scan_agg_hash(tab2, 0, 200_000, [(1, 1, c, 0)],
                             [(0, 0), (1, 1), (2, 2)])
hs()
t0 = time.time(); n = 3
for _ in range(n): hs()
dt = (time.time() - t0) / n
print(f"hash-sparse-100k-groups: {dt*1000:.2f} ms/pass = "
      f"{rows/dt/1e9:.1f}G rows/s", flush=True)
