"""Scan kernel probe: min step time with codec/agg-shape ablation flags.

Usage: probe_scan.py [--rows N] [--codec raw|for] [--aggs full|cnt_si|cnt]
                     [--sel F]
The agg-shape ablations bound which resource the FoR walker is spending
its time on (f32 gather vs agg machinery vs extraction).
"""
import argparse
import ctypes as CT
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

ap = argparse.ArgumentParser()
ap.add_argument("--rows", type=int, default=400_000_000)
ap.add_argument("--codec", default="raw", choices=["raw", "for"])
ap.add_argument("--aggs", default="full", choices=["full", "cnt_si", "cnt"])
ap.add_argument("--sel", type=float, default=0.1)
args = ap.parse_args()

rows, ngroups = args.rows, 1024
rng = np.random.default_rng(44)
keys = rng.integers(0, ngroups, rows).astype(np.int64)
v1 = rng.integers(0, 1 << 20, rows).astype(np.int64)
v2 = rng.normal(0, 1, rows).astype(np.float32)
ctx = sa.GpuContext(0)
lib = sa.gpu()


class ColView(CT.Structure):
    _fields_ = [("data", CT.c_void_p), ("rows", CT.c_uint64), ("type", CT.c_int)]


class PredSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int), ("ilo", CT.c_int64),
                ("ihi", CT.c_int64), ("flo", CT.c_float), ("fhi", CT.c_float)]


class AggSpec(CT.Structure):
    _fields_ = [("col", CT.c_uint32), ("op", CT.c_int)]


class AggResult(CT.Structure):
    _fields_ = [("i64", CT.c_int64), ("f64", CT.c_double)]


if args.codec == "for":
    kb = np.frombuffer(sa.encode_col_i64(keys), dtype=np.uint8)
    vb = np.frombuffer(sa.encode_col_i64(v1), dtype=np.uint8)
    cols = (ColView * 3)(
        ColView(kb.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(vb.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
else:
    cols = (ColView * 3)(
        ColView(keys.ctypes.data_as(CT.c_void_p).value, rows, 0),
        ColView(v1.ctypes.data_as(CT.c_void_p).value, rows, 0),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
tab = CT.c_void_p(0)
assert lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(rows),
                              CT.byref(tab)) == 0
preds = (PredSpec * 1)(
    PredSpec(1, 1, int((1 << 20) * args.sel), 0, 0, 0))
shapes = {
    "full": [(0, 0), (1, 1), (2, 2)],
    "cnt_si": [(0, 0), (1, 1)],
    "cnt": [(0, 0)],
}[args.aggs]
naggs = len(shapes)
aggs = (AggSpec * naggs)(*[AggSpec(c, o) for c, o in shapes])
out = (AggResult * (ngroups * naggs))()
passed = CT.c_uint64(0)


def step():
    assert lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs,
                                naggs, out, CT.byref(passed)) == 0


for _ in range(3):
    step()
best = 1e9
for _ in range(8):
    t0 = time.time()
    step()
    best = min(best, time.time() - t0)
print(f"codec={args.codec} aggs={args.aggs} sel={args.sel} "
      f"nostage={bool(os.environ.get('SDB_SCAN_NOSTAGE'))}: "
      f"best step {best*1000:.2f}ms -> {rows/best/1e9:.1f}G rows/s")
