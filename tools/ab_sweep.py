#!/usr/bin/env python3
"""Same-box A/B of the lean sweep-kernel geometries (SDB_SWEEP_GEOM is read
per execute call, so one resident corpus serves every variant).

  python tools/ab_sweep.py [--docs 100000000] [--steps 10] [--warmup 2]
         [--geoms 0,24576x1024,...] [--k 1000] [--wand]

Prints one line per geometry: ms/step, kernel ms, postings/s, and asserts
hit-set equality against the first variant (parity across geometries).
"""

import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa  # noqa: E402

DEFAULT_GEOMS = "0,24576x1024,16384x1024,12288x512,8192x512,8192x256,4096x256"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=100_000_000)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--k", type=int, default=1000)
    ap.add_argument("--geoms", default=DEFAULT_GEOMS)
    ap.add_argument("--wand", action="store_true")
    ap.add_argument("--libs", default="",
                    help="name=path,... : extra GPU libs to compare on the "
                         "same corpus (fresh context + segment reload per "
                         "lib; ablation/timing builds)")
    args = ap.parse_args()

    sels = [0.10, 0.05, 0.02, 0.01]
    t0 = time.time()
    blob = sa.build_synth_segment(43, 1, args.docs, sels)
    print(f"build {time.time()-t0:.1f}s blob {len(blob)/1e6:.0f}MB",
          flush=True)
    term_idx = [0, 1, 2, 3]
    boosts = [1.0] * 4

    import ctypes as CT
    ref = None
    postings = None
    libspecs = [("default", None)]
    for ent in (args.libs.split(",") if args.libs else []):
        nm, pth = ent.split("=", 1)
        libspecs.append((nm, pth))
    for libname, libpath in libspecs:
      if libpath is None:
          # leave an externally exported SDB_GPU_LIB alone unless this
          # run itself is comparing libs
          if len(libspecs) > 1:
              os.environ.pop("SDB_GPU_LIB", None)
      else:
          os.environ["SDB_GPU_LIB"] = libpath
      sa._gpu = None  # re-resolve the GPU library for this arm
      ctx = sa.GpuContext(0)
      seg = ctx.load_segment(blob)
      lib = sa.gpu()
      for geom in args.geoms.split(","):
          if geom == "wave":
              os.environ.pop("SDB_SWEEP_GEOM", None)
              os.environ["SDB_TOPK_PATH"] = "wave"
          else:
              os.environ.pop("SDB_TOPK_PATH", None)
              os.environ["SDB_SWEEP_GEOM"] = geom
          for _ in range(args.warmup):
              hits, total = ctx.execute_topk([seg], term_idx, boosts, args.k,
                                             wand=args.wand)
          kms = 0.0
          t1 = time.time()
          for _ in range(args.steps):
              hits, total = ctx.execute_topk([seg], term_idx, boosts, args.k,
                                             wand=args.wand)
              ms = CT.c_double(0)
              lib.sdb_gpu_last_kernel_ms(ctx._ctx, CT.byref(ms))
              kms += ms.value
          el = time.time() - t1
          if postings is None:
              # postings per query = sum of term dfs (parse once)
              host = sa.host()
              v = np.frombuffer(blob, dtype=np.uint8)

              class _View(CT.Structure):
                  _fields_ = [("hdr", CT.c_void_p), ("terms", CT.c_void_p),
                              ("desc", CT.c_void_p), ("norms", CT.c_void_p),
                              ("payload", CT.c_void_p)]

              class _Term(CT.Structure):
                  _fields_ = [("desc_begin", CT.c_uint64),
                              ("desc_end", CT.c_uint64),
                              ("payload_begin", CT.c_uint64),
                              ("payload_end", CT.c_uint64),
                              ("df", CT.c_uint32), ("max_freq", CT.c_uint32),
                              ("total_freq", CT.c_uint64)]
              vw = _View()
              host.sdb_host_segment_parse(v.ctypes.data_as(CT.c_void_p),
                                          CT.c_uint64(len(v)), CT.byref(vw))
              terms = CT.cast(vw.terms, CT.POINTER(_Term * 4)).contents
              postings = sum(terms[t].df for t in range(4))
          cur = (hits["doc"].copy(), hits["score"].view(np.uint32).copy(),
                 total)
          if ref is None:
              ref = cur
              ok = "ref"
          else:
              ok = ("OK" if (np.array_equal(ref[0], cur[0]) and
                             np.array_equal(ref[1], cur[1]) and
                             ref[2] == cur[2]) else "MISMATCH")
          print(f"lib={libname:8s} geom={geom:12s} "
                f"{el*1000/args.steps:8.3f} ms/step "
                f"kernel {kms/args.steps:7.3f} ms  "
                f"{postings*args.steps/el/1e9:7.2f}G postings/s  parity={ok}",
                flush=True)


if __name__ == "__main__":
    main()
