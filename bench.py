#!/usr/bin/env python3
"""bench.py — headline benchmark of the MI355X-native SereneDB hot path.

Default workload = "all": the FULL BASELINE metric ("docs scored/sec BM25
top-1000 (+ rows/sec filter-agg)") — the primary line is configs[1] (BM25
top-1000, 4-term disjunction, 100M synthetic docs, seed 43, SURVEY.md §8d
distributions) and the same JSON line embeds the configs[2] scan
(1B-row scan->filter->group-by, seed 44) and configs[3] hybrid results
under "extra_benches", each with its own roofline and cpu_baseline (the
round-1 driver record certified only the BM25 half — VERDICT #4).
A "step" is one full query execution (decode -> score -> top-k -> merge),
resp. one full scan->filter->group-by pass.

  python bench.py --gpus N --steps K --warmup W [--workload bm25_topk]

For N>1 the driver launches this under torch.distributed.run (one rank per
GPU over RCCL); ranks shard the SAME corpus by doc range (strong scaling,
BASELINE configs[4]) and merge per-rank top-k candidates with an allgather
(24 KB — SURVEY.md §8e) plus an allreduce of total-match counts.

Outputs ONE JSON line from rank 0 (driver contract).

roofline.traffic: HBM bytes per launch from rocprofv3 PMC counters
(FETCH_SIZE/WRITE_SIZE with the gfx950 unit correction calibrated on a
known-byte-count pattern — tools/pmc_traffic.py), read from the committed
profiles/pmc_traffic.json when it matches this workload/config; null
otherwise.
"""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK_GBS = 8000.0  # 8 TB/s spec (MI355X_MICROARCH.md)


def env_rank():
    return (int(os.environ.get("RANK", "0")),
            int(os.environ.get("WORLD_SIZE", "1")))


def pmc_traffic(workload_key):
    """Per-launch HBM traffic for the named workload, from the committed
    PMC measurement (tools/pmc_traffic.py -> profiles/pmc_traffic.json).
    Returns (bytes_per_launch, note) or (None, None)."""
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "profiles", "pmc_traffic.json")
    try:
        with open(path) as f:
            data = json.load(f)
    except Exception:
        return None, None
    ent = data.get(workload_key)
    if not ent:
        return None, None
    return ent.get("bytes_per_launch"), ent.get("note")


def dist_init(world):
    if world <= 1:
        return None
    import torch
    import torch.distributed as tdist

    tdist.init_process_group(backend="nccl")
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return tdist


def pack_hits(hits, base, k):
    """(score,doc) -> int64 [k]: score_bits<<32 | (2^32-1 - global_doc), so a
    DESCENDING sort of packed values orders by (score desc, doc asc) — the
    deterministic tie order of DESIGN.md. Scores >= 0 so float order ==
    bit order; padding is -1 (negative, filtered out)."""
    out = np.full(k, -1, dtype=np.int64)
    n = len(hits)
    sb = hits["score"].view(np.uint32).astype(np.int64)
    gd = hits["doc"].astype(np.int64) + base
    out[:n] = (sb << 32) | (0xFFFFFFFF - gd)
    return out


def unpack_hits(packed):
    packed = packed[packed >= 0]
    sb = (packed >> 32).astype(np.uint32)
    docs = (0xFFFFFFFF - (packed & 0xFFFFFFFF)).astype(np.uint64)
    scores = sb.view(np.float32)
    return scores, docs


def bench_bm25(args, dist, hybrid=False):
    import serenedb_amd as sa

    rank, world = env_rank()
    seed = 43
    doc_count = args.docs
    sels = [0.10, 0.05, 0.02, 0.01]
    k = 1000
    nterms = len(sels)

    # ---- shard build (untimed; index-build side is CPU by design) ----
    per = doc_count // world
    lo = rank * per + 1
    hi = doc_count if rank == world - 1 else (rank + 1) * per
    t0 = time.time()
    blob = sa.build_synth_segment(seed, lo, hi, sels)
    build_s = time.time() - t0

    device = int(os.environ.get("LOCAL_RANK", 0))
    ctx = sa.GpuContext(device)
    seg = ctx.load_segment(blob)

    # hybrid (configs[3]): i64 filter column over this shard's docs,
    # BETWEEN at 20% selectivity, 64-bucket COUNT/SUM group-by (seed 45)
    nbuckets = 64
    span = 1 << 31
    flo, fhi = int(span * 0.4), int(span * 0.6) - 1
    if hybrid:
        crng = np.random.default_rng(45 + rank)
        col = crng.integers(0, span, (hi - lo + 1) + 1).astype(np.int64)
        ctx.attach_column(seg, col)

    # ---- global stats (PreparePhase analogue over RCCL) ----
    import ctypes as CT
    v = np.frombuffer(blob, dtype=np.uint8)
    host = sa.host()

    class _View(CT.Structure):
        _fields_ = [("hdr", CT.c_void_p), ("terms", CT.c_void_p),
                    ("desc", CT.c_void_p), ("norms", CT.c_void_p),
                    ("payload", CT.c_void_p)]

    class _Hdr(CT.Structure):
        _fields_ = [("magic", CT.c_uint64), ("version", CT.c_uint32),
                    ("nterms", CT.c_uint32), ("doc_count", CT.c_uint32),
                    ("docs_with_field", CT.c_uint32),
                    ("total_term_freq", CT.c_uint64),
                    ("total_blocks", CT.c_uint64),
                    ("off_terms", CT.c_uint64), ("off_desc", CT.c_uint64),
                    ("off_norms", CT.c_uint64), ("off_payload", CT.c_uint64),
                    ("payload_size", CT.c_uint64),
                    ("blob_size", CT.c_uint64)]

    class _Term(CT.Structure):
        _fields_ = [("desc_begin", CT.c_uint64), ("desc_end", CT.c_uint64),
                    ("payload_begin", CT.c_uint64),
                    ("payload_end", CT.c_uint64), ("df", CT.c_uint32),
                    ("max_freq", CT.c_uint32), ("total_freq", CT.c_uint64)]

    vw = _View()
    rc = host.sdb_host_segment_parse(
        v.ctypes.data_as(CT.c_void_p), CT.c_uint64(len(v)), CT.byref(vw))
    assert rc == 0
    h = CT.cast(vw.hdr, CT.POINTER(_Hdr)).contents
    terms = CT.cast(vw.terms, CT.POINTER(_Term * h.nterms)).contents
    local_stats = np.array(
        [h.docs_with_field, h.total_term_freq] +
        [terms[t].df for t in range(nterms)], dtype=np.int64)
    local_payload_bytes = sum(
        terms[t].payload_end - terms[t].payload_begin for t in range(nterms))
    local_desc_blocks = sum(
        terms[t].desc_end - terms[t].desc_begin for t in range(nterms))
    if dist:
        import torch
        ts = torch.tensor(local_stats, device="cuda")
        dist.all_reduce(ts)  # PreparePhase stats merge over RCCL
        gstats_arr = ts.cpu().numpy()
    else:
        gstats_arr = local_stats
    g_dwf, g_ttf = int(gstats_arr[0]), int(gstats_arr[1])
    g_dwt = [int(x) for x in gstats_arr[2:2 + nterms]]
    gstats = (g_dwf, g_ttf, g_dwt)
    total_postings_global = sum(g_dwt)
    term_idx = list(range(nterms))
    boosts = [1.0] * nterms

    lib = sa.gpu()
    lib.sdb_gpu_last_kernel_ms.restype = CT.c_int

    def step():
        if hybrid:
            hits, total, bcnt, bsum = ctx.execute_topk_hybrid(
                [seg], term_idx, boosts, k, flo, fhi, nbuckets,
                global_stats=gstats)
            if dist:
                import torch
                bb = torch.from_numpy(np.concatenate([bcnt, bsum])).cuda()
                dist.all_reduce(bb)  # partial-agg merge
                _ = bb.cpu().numpy()
        else:
            hits, total = ctx.execute_topk([seg], term_idx, boosts, k,
                                           global_stats=gstats)
        if dist:
            import torch
            packed = torch.from_numpy(pack_hits(hits, lo - 1, k)).cuda()
            gathered = [torch.empty_like(packed) for _ in range(world)]
            dist.all_gather(gathered, packed)
            tm = torch.tensor([total], dtype=torch.int64, device="cuda")
            dist.all_reduce(tm)
            total = int(tm.item())
            allp = torch.cat(gathered).cpu().numpy()
            allp.sort()
            topk = allp[::-1][:k]  # score desc; doc tiebreak implicit in bits
            scores, docs = unpack_hits(topk)
            return scores, docs, total
        return hits["score"], hits["doc"], total

    def sync():
        if dist:
            import torch
            dist.barrier()
            torch.cuda.synchronize()

    # ---- warmup ----
    for _ in range(args.warmup):
        step()
    sync()

    # ---- timed ----
    # Default execution = the pipelined batch entry
    # (sdb_gpu_execute_topk_batch): every query fully re-executed (memset
    # -> kernels -> readback -> exact select), query k+1's kernels enqueue
    # while query k's candidates read back and select on the host — the
    # production QPS shape. Kernel-trace cadence 0.49 ms/query vs 0.67
    # per-step at 100M (profiles/r2_evidence/r2_batch_trace.log); results
    # bit-equal per query (tests/test_gpu_parity.py
    # test_batch_pipelined_equals_single, re-asserted in warmup here).
    # Hybrid keeps per-step (the batch entry carries no bucket state).
    use_batch = args.execution == "batch"
    if use_batch and not hybrid:
        ref_hits, ref_total = ctx.execute_topk([seg], term_idx, boosts, k,
                                               global_stats=gstats)
        bh, bt = ctx.execute_topk_batch([seg], term_idx, boosts, k, 2,
                                        global_stats=gstats, all_hits=True)
        assert bt[0] == ref_total and len(bh[0]) == len(ref_hits)
        assert all(tuple(a) == tuple(b) for a, b in zip(bh[0], ref_hits))
    elif use_batch:
        rh, rt, rc_, rs = ctx.execute_topk_hybrid(
            [seg], term_idx, boosts, k, flo, fhi, nbuckets,
            global_stats=gstats)
        bh, bt, bc, bs = ctx.execute_topk_hybrid_batch(
            [seg], term_idx, boosts, k, flo, fhi, nbuckets, 2,
            global_stats=gstats, all_hits=True)
        assert bt[0] == rt and len(bh[0]) == len(rh)
        assert all(tuple(a) == tuple(b) for a, b in zip(bh[0], rh))
        assert (bc[0] == rc_).all() and (bs[0] == rs).all()
    kernel_ms_acc = 0.0
    sync()
    t0 = time.time()
    if use_batch:
        if hybrid:
            hits_l, totals_l, bc, bs = ctx.execute_topk_hybrid_batch(
                [seg], term_idx, boosts, k, flo, fhi, nbuckets,
                args.steps, global_stats=gstats, all_hits=bool(dist))
        else:
            hits_l, totals_l = ctx.execute_topk_batch(
                [seg], term_idx, boosts, k, args.steps,
                global_stats=gstats, all_hits=bool(dist))
        ms = CT.c_double(0)
        lib.sdb_gpu_last_kernel_ms(ctx._ctx, CT.byref(ms))
        kernel_ms_acc = ms.value * args.steps  # batch reports total/nq
        total = totals_l[-1]
        if dist:
            import torch
            for q in range(args.steps):
                packed = torch.from_numpy(
                    pack_hits(hits_l[q], lo - 1, k)).cuda()
                gathered = [torch.empty_like(packed) for _ in range(world)]
                dist.all_gather(gathered, packed)
                tm = torch.tensor([totals_l[q]], dtype=torch.int64,
                                  device="cuda")
                dist.all_reduce(tm)
                total = int(tm.item())
                if hybrid:
                    bb = torch.from_numpy(
                        np.concatenate([bc[q], bs[q]])).cuda()
                    dist.all_reduce(bb)  # partial-agg merge
                    _ = bb.cpu().numpy()
    else:
        for _ in range(args.steps):
            scores, docs, total = step()
            ms = CT.c_double(0)
            lib.sdb_gpu_last_kernel_ms(ctx._ctx, CT.byref(ms))
            kernel_ms_acc += ms.value
    sync()
    elapsed = time.time() - t0
    if dist:
        import torch
        te = torch.tensor([elapsed], device="cuda")
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())

    ms_per_step = elapsed * 1000.0 / args.steps
    # BASELINE's metric is DOCS scored/sec: every matched doc is scored
    # exactly once (round-1 weak #5 was a postings/s value under a docs/s
    # label); postings/s kept in config for continuity
    value = total * args.steps / elapsed
    postings_per_sec = total_postings_global * args.steps / elapsed

    # ---- roofline (rank 0's shard kernel): algorithmic bytes per launch ----
    # postings payload (compressed docs+freqs+embedded norm blocks, tag
    # bytes included — format v2 materializes the norm-column gather into
    # the payload stream) + 28 B descriptor per touched block
    local_postings = sum(terms[t].df for t in range(nterms))
    algo_bytes = local_payload_bytes + 28 * local_desc_blocks
    kernel_s = kernel_ms_acc / 1000.0
    achieved_gbs = (algo_bytes * args.steps / kernel_s / 1e9) if kernel_s else 0
    wl_key = (("hybrid_" if hybrid else "") +
              ("bm25_top1000_4term_or_100M" if doc_count == 100_000_000
               else f"bm25_top1000_4term_or_{doc_count}"))
    traffic, traffic_note = pmc_traffic(wl_key)
    roofline = {
        "bound": "hbm",
        "achieved": round(achieved_gbs, 1),
        "peak": HBM_PEAK_GBS,
        "unit": "GB/s",
        "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
        "traffic": traffic,
        "note": "achieved = algorithmic bytes (compressed postings payload incl. per-block doc/freq/norm streams + 28B descriptor per block) / window-kernel time (HIP events on the library stream); rocprofv3 evidence under profiles/"
                + ("; traffic = PMC FETCH+WRITE per launch, " + traffic_note
                   if traffic else ""),
    }

    # ---- CPU baseline (rank 0, N=1 only): the oracle's multithreaded
    # mechanics path (RunTopKScan restatement) on a bounded sample ----
    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        from oracle import pyoracle as po

        # the full corpus IS the sample at N=1 (most honest: identical
        # workload; per-query CPU time ~5-15 ms so a 10 s run is bounded);
        # hybrid's exact-path oracle materializes candidates, keep it on a
        # 12.5M shard
        sample_docs = doc_count if not hybrid else min(doc_count, 12_500_000)
        sblob = (blob if (lo == 1 and hi == sample_docs) else
                 sa.build_synth_segment(seed, 1, sample_docs, sels))
        # postings in the sample
        sv = _View()
        sb = np.frombuffer(sblob, dtype=np.uint8)
        host.sdb_host_segment_parse(sb.ctypes.data_as(CT.c_void_p),
                                    CT.c_uint64(len(sb)), CT.byref(sv))
        sh = CT.cast(sv.hdr, CT.POINTER(_Hdr)).contents
        st = CT.cast(sv.terms, CT.POINTER(_Term * sh.nterms)).contents
        sample_postings = sum(st[t].df for t in range(nterms))
        ncores = os.cpu_count() or 1
        iters = 0
        scol = None
        if hybrid:
            crng = np.random.default_rng(45)
            scol = crng.integers(0, span, sample_docs + 1).astype(np.int64)
        if hybrid:
            tcpu = time.time()
            while time.time() - tcpu < args.cpu_seconds and iters < 200:
                po.execute_topk_hybrid(sblob, term_idx, boosts, k, scol,
                                       flo, fhi, nbuckets,
                                       global_stats=gstats)
                iters += 1
        else:
            # estimate per-query time, then run the whole workload inside
            # the thread pool (amortizes 256 pthread spawns per query)
            t1 = time.time()
            po.execute_topk_mt(sblob, term_idx, boosts, k, nthreads=ncores,
                               global_stats=gstats, iters=4)
            est = (time.time() - t1) / 4
            iters = max(1, min(600, int(args.cpu_seconds / max(est, 1e-4))))
            tcpu = time.time()
            po.execute_topk_mt(sblob, term_idx, boosts, k, nthreads=ncores,
                               global_stats=gstats, iters=iters)
        tcpu = time.time() - tcpu
        cpu_baseline = {
            "value": round(sample_postings * iters / tcpu, 1),
            "unit": "postings scored/s",  # compare with
                                          # config.postings_per_sec
            "cores": 1 if hybrid else ncores,
            "kind": "port",
            "sample": f"{sample_docs/1e6:.1f}M-doc shard of the same corpus "
                      f"({sample_postings} postings/query, {iters} iters, "
                      f"{tcpu:.1f}s; oracle "
                      + ("hybrid exact path, single-thread"
                         if hybrid else "RunTopKScan restatement")
                      + ")",
        }

    result = {
        "metric": ("docs scored/sec BM25 top-1000 + range filter-agg"
                   if hybrid else "docs scored/sec BM25 top-1000"),
        "value": round(value, 1),
        "unit": "docs/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "f32",
        "data": "synthetic",
        "config": {
            "workload": (("hybrid_" if hybrid else "") +
                         ("bm25_top1000_4term_or_100M"
                          if doc_count == 100_000_000
                          else f"bm25_top1000_4term_or_{doc_count}")),
            "execution": "pipelined-batch" if use_batch else "per-step",
            "doc_count": doc_count,
            "selectivities": sels,
            "k": k,
            "seed": seed,
            "postings_per_query": int(total_postings_global),
            "postings_per_sec": round(postings_per_sec, 1),
            "total_matches": int(total),
            "parallelism": f"doc-range shards x{world}, RCCL allgather merge",
            "segment_build_s": round(build_s, 1),
        },
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    ctx.close()
    return result


def bench_scan(args, dist):
    import ctypes as CT

    import serenedb_amd as sa

    rank, world = env_rank()
    rows_total = args.rows
    ngroups = 1024
    seed = 44
    per = rows_total // world
    my_rows = per if rank != world - 1 else rows_total - per * (world - 1)
    rng = np.random.default_rng(seed + rank)
    keys = rng.integers(0, ngroups, my_rows).astype(np.int64)
    v1 = rng.integers(0, 1 << 20, my_rows).astype(np.int64)
    v2 = rng.normal(0, 1, my_rows).astype(np.float32)
    use_for = args.scan_codec == "for"
    if use_for:
        keys_blob = sa.encode_col_i64(keys)
        v1_blob = sa.encode_col_i64(v1)
        kb = np.frombuffer(keys_blob, dtype=np.uint8)
        vb = np.frombuffer(v1_blob, dtype=np.uint8)
        scan_bytes = len(keys_blob) + len(v1_blob) + 4 * my_rows
    else:
        scan_bytes = 20 * my_rows

    device = int(os.environ.get("LOCAL_RANK", 0))
    ctx = sa.GpuContext(device)
    lib = sa.gpu()

    class ColView(CT.Structure):
        _fields_ = [("data", CT.c_void_p), ("rows", CT.c_uint64),
                    ("type", CT.c_int)]

    class PredSpec(CT.Structure):
        _fields_ = [("col", CT.c_uint32), ("op", CT.c_int),
                    ("ilo", CT.c_int64), ("ihi", CT.c_int64),
                    ("flo", CT.c_float), ("fhi", CT.c_float)]

    class AggSpec(CT.Structure):
        _fields_ = [("col", CT.c_uint32), ("op", CT.c_int)]

    class AggResult(CT.Structure):
        _fields_ = [("i64", CT.c_int64), ("f64", CT.c_double)]

    if use_for:
        cols = (ColView * 3)(
            ColView(kb.ctypes.data_as(CT.c_void_p).value, my_rows, 2),
            ColView(vb.ctypes.data_as(CT.c_void_p).value, my_rows, 2),
            ColView(v2.ctypes.data_as(CT.c_void_p).value, my_rows, 1))
    else:
        cols = (ColView * 3)(
            ColView(keys.ctypes.data_as(CT.c_void_p).value, my_rows, 0),
            ColView(v1.ctypes.data_as(CT.c_void_p).value, my_rows, 0),
            ColView(v2.ctypes.data_as(CT.c_void_p).value, my_rows, 1))
    tab = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(my_rows),
                                CT.byref(tab))
    assert rc == 0, rc
    c = int((1 << 20) * 0.1)  # 10% selectivity predicate
    preds = (PredSpec * 1)(PredSpec(1, 1, c, 0, 0, 0))
    aggs = (AggSpec * 3)(AggSpec(0, 0), AggSpec(1, 1), AggSpec(2, 2))
    out = (AggResult * (ngroups * 3))()
    passed = CT.c_uint64(0)

    def step():
        rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs,
                                  3, out, CT.byref(passed))
        assert rc == 0, rc
        if dist:
            import torch
            # 24 KB partial-agg merge: COUNT/SUM(i64) slots
            # merge in the i64 field, SUM(f32->f64) slots in the f64 field
            arr = np.array([out[i].i64 for i in range(ngroups * 3)],
                           dtype=np.int64)
            farr = np.array([out[i].f64 for i in range(ngroups * 3)],
                            dtype=np.float64)
            t = torch.from_numpy(arr).cuda()
            tf = torch.from_numpy(farr).cuda()
            dist.all_reduce(t)
            dist.all_reduce(tf)
            _ = t.cpu().numpy()
            _ = tf.cpu().numpy()

    def sync():
        if dist:
            import torch
            dist.barrier()
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.time() - t0
    if dist:
        import torch
        te = torch.tensor([elapsed], device="cuda")
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
    ms_per_step = elapsed * 1000 / args.steps
    value = rows_total * args.steps / elapsed

    algo_bytes = scan_bytes  # compressed (FoR) or raw column bytes scanned
    achieved_gbs = algo_bytes * args.steps / elapsed / 1e9  # whole step ~ kernel
    scan_wl_key = (f"scan_filter_groupby_{rows_total//10**9}B"
                   if rows_total >= 10**9
                   else f"scan_filter_groupby_{rows_total}")
    scan_traffic, scan_traffic_note = pmc_traffic(scan_wl_key)
    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        from oracle import pyoracle as po

        ncores = os.cpu_count() or 1
        t1 = time.time()
        po.scan_agg_mt(keys, v1, v2, ngroups, pred_op=1, lo=c,
                       nthreads=ncores, iters=2)
        est = (time.time() - t1) / 2
        iters = max(1, min(300, int(args.cpu_seconds / max(est, 1e-3))))
        t1 = time.time()
        po.scan_agg_mt(keys, v1, v2, ngroups, pred_op=1, lo=c,
                       nthreads=ncores, iters=iters)
        t1 = time.time() - t1
        cpu_baseline = {
            "value": round(my_rows * iters / t1, 1),
            "unit": "rows/s",
            "cores": ncores,
            "kind": "port",
            "sample": f"full {my_rows/1e6:.0f}M-row table, {iters} iters, "
                      f"{t1:.1f}s (oracle scan_agg, {ncores} threads, "
                      "in-pool iteration loop)",
        }
    result = {
        "metric": "rows/sec filter-agg",
        "value": round(value, 1),
        "unit": "rows/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "int64",
        "data": "synthetic",
        "config": {
            "workload": f"scan_filter_groupby_{rows_total//10**9}B" if rows_total >= 10**9 else f"scan_filter_groupby_{rows_total}",
            "rows": rows_total,
            "ngroups": ngroups,
            "predicate": "v1 < 10% quantile",
            "aggs": "COUNT, SUM(i64), SUM(f32->f64)",
            "codec": args.scan_codec,
            "bytes_per_row": round(scan_bytes / my_rows, 2),
            "seed": seed,
            "parallelism": f"row shards x{world}, RCCL allreduce merge",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
            "traffic": scan_traffic,
            "note": ("traffic = PMC FETCH+WRITE per launch, " +
                     scan_traffic_note) if scan_traffic else None,
        },
        "cpu_baseline": cpu_baseline,
    }
    lib.sdb_gpu_table_free(ctx._ctx, tab)
    ctx.close()
    return result


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--workload", default="all",
                    choices=["all", "bm25_topk", "scan_agg", "hybrid"])
    ap.add_argument("--docs", type=int, default=100_000_000)
    ap.add_argument("--rows", type=int, default=1_000_000_000)
    ap.add_argument("--scan-codec", default="raw", choices=["for", "raw"])
    ap.add_argument("--execution", default="batch", choices=["batch", "step"])
    ap.add_argument("--cpu-seconds", type=float, default=10.0)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()
    rank, world = env_rank()
    dist = dist_init(world)

    if args.workload == "bm25_topk":
        primary = bench_bm25(args, dist, hybrid=False)
    elif args.workload == "hybrid":
        primary = bench_bm25(args, dist, hybrid=True)
    elif args.workload == "scan_agg":
        primary = bench_scan(args, dist)
    else:
        # the full BASELINE metric: BM25 primary + scan and hybrid
        # embedded, every half driver-certified in one record
        primary = bench_bm25(args, dist, hybrid=False)
        scan = bench_scan(args, dist)
        hyb = bench_bm25(args, dist, hybrid=True)
        primary["metric"] = \
            "docs scored/sec BM25 top-1000 (+ rows/sec filter-agg)"
        primary["extra_benches"] = [scan, hyb]
    if rank == 0:
        print(json.dumps(primary), flush=True)
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
