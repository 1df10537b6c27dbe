#!/usr/bin/env python3
"""Batch vs per-step top-k execution probe (the ROUND3_NOTES post-mortem:
batch measured +25-35% ms/step at 100M — diagnose before reuse).

  python tools/probe_batch.py [--docs N] [--nq 30] [--mode both|step|batch]

--mode step/batch runs ONE mode only (for separate rocprofv3 traces).
Prints ms/step and the library's own kernel-ms accounting per mode.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=100_000_000)
    ap.add_argument("--nq", type=int, default=30)
    ap.add_argument("--k", type=int, default=1000)
    ap.add_argument("--mode", default="both",
                    choices=["both", "step", "batch"])
    args = ap.parse_args()

    t0 = time.time()
    blob = sa.build_synth_segment(43, 1, args.docs, [0.10, 0.05, 0.02, 0.01])
    ctx = sa.GpuContext(0)
    seg = ctx.load_segment(blob)
    print(f"setup {time.time()-t0:.1f}s", flush=True)
    term_idx = [0, 1, 2, 3]
    boosts = [1.0] * 4

    import ctypes as CT

    def kms():
        ms = CT.c_double(0)
        ctx._lib.sdb_gpu_last_kernel_ms(ctx._ctx, CT.byref(ms))
        return ms.value

    ref = None
    if args.mode in ("both", "step"):
        for _ in range(3):
            ref, _tot = ctx.execute_topk([seg], term_idx, boosts, args.k)
        t0 = time.time()
        for _ in range(args.nq):
            ctx.execute_topk([seg], term_idx, boosts, args.k)
        dt = (time.time() - t0) / args.nq
        print(f"per-step: {dt*1e3:.3f} ms/step kernel {kms():.3f} ms",
              flush=True)

    if args.mode in ("both", "batch"):
        ctx.execute_topk_batch([seg], term_idx, boosts, args.k, 4)
        t0 = time.time()
        hits, totals = ctx.execute_topk_batch(
            [seg], term_idx, boosts, args.k, args.nq, all_hits=True)
        dt = (time.time() - t0) / args.nq
        print(f"batch:    {dt*1e3:.3f} ms/step kernel(total/nq) {kms():.3f} ms",
              flush=True)
        if ref is not None:
            h0 = hits[0]
            assert len(h0) == len(ref), (len(h0), len(ref))
            for a, b in zip(h0, ref):
                assert tuple(a) == tuple(b), (a, b)
            print("parity: batch q0 == per-step (bit-exact)")


if __name__ == "__main__":
    main()
