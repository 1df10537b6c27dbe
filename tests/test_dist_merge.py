"""Multi-process (gloo, world_size=2) test of the distributed merge path on
CPU: per-rank shard execution (oracle standing in for the per-GPU kernel) +
torch.distributed stats-allreduce + top-k allgather + host merge must equal
the single-segment result. This covers bench.py's N>1 logic (pack/unpack,
stats merge, candidate merge) without a GPU.
"""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

DOCS = 60_000
SELS = [0.05, 0.02, 0.01]
K = 120
SEED = 13


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import serenedb_amd as sa
    from oracle import pyoracle as po
    import bench

    per = DOCS // world
    lo = rank * per + 1
    hi = DOCS if rank == world - 1 else (rank + 1) * per
    blob = sa.build_synth_segment(SEED, lo, hi, SELS)

    # stats allreduce (PreparePhase analogue)
    postings = [sa.synth_postings(SEED, DOCS, t, s)[0][
        (sa.synth_postings(SEED, DOCS, t, s)[0] >= lo) &
        (sa.synth_postings(SEED, DOCS, t, s)[0] <= hi)]
        for t, s in enumerate(SELS)]
    norms = sa.synth_norms(SEED, DOCS)
    local = torch.tensor(
        [hi - lo + 1, int(norms[lo:hi + 1].sum())] +
        [len(p) for p in postings], dtype=torch.int64)
    dist.all_reduce(local)
    g_dwf, g_ttf = int(local[0]), int(local[1])
    g_dwt = [int(x) for x in local[2:]]

    hits, total = po.execute_topk([blob], list(range(len(SELS))),
                                  [1.0] * len(SELS), K,
                                  global_stats=(g_dwf, g_ttf, g_dwt))
    packed = torch.from_numpy(bench.pack_hits(hits, lo - 1, K))
    gathered = [torch.empty_like(packed) for _ in range(world)]
    dist.all_gather(gathered, packed)
    tm = torch.tensor([total], dtype=torch.int64)
    dist.all_reduce(tm)

    if rank == 0:
        allp = torch.cat(gathered).numpy()
        allp = np.sort(allp)[::-1][:K]
        scores, docs = bench.unpack_hits(allp)
        q.put((scores.copy(), docs.copy(), int(tm.item())))
    dist.destroy_process_group()


def _scan_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from oracle import pyoracle as po

    rows_total, ngroups, seed = 120_000, 64, 21
    per = rows_total // world
    lo = rank * per
    hi = rows_total if rank == world - 1 else (rank + 1) * per
    rng = np.random.default_rng(seed)  # same stream; slice the shard
    keys = rng.integers(0, ngroups, rows_total).astype(np.int64)[lo:hi]
    v1r = rng.integers(0, 1 << 20, rows_total).astype(np.int64)
    v2r = rng.normal(0, 1, rows_total).astype(np.float32)
    v1, v2 = v1r[lo:hi], v2r[lo:hi]
    c = int((1 << 20) * 0.1)
    cnt, si, sf, passed = po.scan_agg(keys, v1, v2, ngroups, pred_op=1,
                                      lo=c)
    ti = torch.from_numpy(np.concatenate(
        [cnt.astype(np.int64), si.astype(np.int64), [np.int64(passed)]]))
    tf = torch.from_numpy(sf.astype(np.float64))
    dist.all_reduce(ti)
    dist.all_reduce(tf)
    if rank == 0:
        q.put((ti.numpy().copy(), tf.numpy().copy()))
    dist.destroy_process_group()


def test_gloo_scan_agg_merge_equals_full():
    """bench.py's N>1 scan-aggregate merge (sum-allreduce of the group
    table, i64 and f64 planes) equals the single-shard result."""
    from oracle import pyoracle as po

    rows_total, ngroups, seed = 120_000, 64, 21
    rng = np.random.default_rng(seed)
    keys = rng.integers(0, ngroups, rows_total).astype(np.int64)
    v1 = rng.integers(0, 1 << 20, rows_total).astype(np.int64)
    v2 = rng.normal(0, 1, rows_total).astype(np.float32)
    c = int((1 << 20) * 0.1)
    fcnt, fsi, fsf, fpassed = po.scan_agg(keys, v1, v2, ngroups, pred_op=1,
                                          lo=c)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_scan_worker, args=(r, 2, 29523, q))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        ti, tf = q.get(timeout=300)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    np.testing.assert_array_equal(ti[:ngroups], fcnt)
    np.testing.assert_array_equal(ti[ngroups:2 * ngroups], fsi)
    assert ti[-1] == fpassed
    np.testing.assert_allclose(tf, fsf, rtol=1e-12)


@pytest.mark.parametrize("world,port", [(2, 29511), (4, 29517)])
def test_gloo_sharded_merge_equals_full(world, port):
    import serenedb_amd as sa
    from oracle import pyoracle as po

    full = sa.build_synth_segment(SEED, 1, DOCS, SELS)
    fhits, ftotal = po.execute_topk([full], list(range(len(SELS))),
                                    [1.0] * len(SELS), K)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    try:
        scores, docs, total = q.get(timeout=300)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    assert total == ftotal
    np.testing.assert_array_equal(docs.astype(np.uint32), fhits["doc"])
    np.testing.assert_array_equal(scores, fhits["score"])


def _hash_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    rows_total, seed = 90_000, 33
    per = rows_total // world
    lo = rank * per
    hi = rows_total if rank == world - 1 else (rank + 1) * per
    rng = np.random.default_rng(seed)
    base = rng.integers(-(1 << 60), 1 << 60, 5000).astype(np.int64)
    keys = base[rng.integers(0, len(base), rows_total)][lo:hi]
    vals = rng.integers(-1000, 1000, rows_total).astype(np.int64)[lo:hi]
    # per-rank "hash aggregate" rows (what sdb_gpu_scan_agg_hash returns:
    # keys ascending + per-key aggregates)
    uk, inv = np.unique(keys, return_inverse=True)
    cnt = np.bincount(inv, minlength=len(uk)).astype(np.int64)
    s = np.zeros(len(uk), dtype=np.int64)
    np.add.at(s, inv, vals)
    # the N>1 merge: variable-size all_gather (gloo object collective),
    # concat + groupby on every rank — the SURVEY §8e "gather + host
    # merge" path for open hash tables
    rows = [None] * world
    dist.all_gather_object(rows, (uk, cnt, s))
    allk = np.concatenate([r[0] for r in rows])
    allc = np.concatenate([r[1] for r in rows])
    alls = np.concatenate([r[2] for r in rows])
    mk, minv = np.unique(allk, return_inverse=True)
    mc = np.zeros(len(mk), dtype=np.int64)
    ms = np.zeros(len(mk), dtype=np.int64)
    np.add.at(mc, minv, allc)
    np.add.at(ms, minv, alls)
    if rank == 0:
        q.put((mk, mc, ms))
    dist.destroy_process_group()


def _strkey_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    from oracle.pyoracle import fnv1a64

    rows_total, seed = 60_000, 35
    per = rows_total // world
    lo = rank * per
    hi = rows_total if rank == world - 1 else (rank + 1) * per
    rng = np.random.default_rng(seed)
    vocab = [f"term-{i:04d}" for i in range(800)]
    vals_s = [vocab[i] for i in rng.integers(0, len(vocab), rows_total)]
    vals = rng.integers(-1000, 1000, rows_total).astype(np.int64)
    # per-rank string-key hash aggregate (what scan_agg_hash_str returns:
    # FNV-1a hash keys signed-ascending + per-key aggregates); ranks
    # shard rows, the merge is identical to the i64 hash-agg merge
    # because the keys ARE i64 hashes
    keys = np.array([np.int64(np.uint64(fnv1a64(v.encode())))
                     for v in vals_s[lo:hi]])
    mv = vals[lo:hi]
    uk, inv = np.unique(keys, return_inverse=True)
    cnt = np.bincount(inv, minlength=len(uk)).astype(np.int64)
    sm = np.zeros(len(uk), dtype=np.int64)
    np.add.at(sm, inv, mv)
    rows = [None] * world
    dist.all_gather_object(rows, (uk, cnt, sm))
    allk = np.concatenate([r[0] for r in rows])
    allc = np.concatenate([r[1] for r in rows])
    alls = np.concatenate([r[2] for r in rows])
    mk, minv = np.unique(allk, return_inverse=True)
    mc = np.zeros(len(mk), dtype=np.int64)
    ms = np.zeros(len(mk), dtype=np.int64)
    np.add.at(mc, minv, allc)
    np.add.at(ms, minv, alls)
    if rank == 0:
        q.put((mk, mc, ms))
    dist.destroy_process_group()


def test_gloo_strkey_groupby_merge_equals_full():
    """Distributed merge of per-rank STRING-key group-bys: the keys are
    FNV-1a 64 hashes (what sdb_gpu_scan_agg_hash_str returns), so the
    merge is the same gather + merge-by-key as the i64 hash aggregate;
    equals the single-node string aggregation resolved back through the
    hash."""
    from oracle.pyoracle import fnv1a64

    rows_total, seed = 60_000, 35
    rng = np.random.default_rng(seed)
    vocab = [f"term-{i:04d}" for i in range(800)]
    vals_s = [vocab[i] for i in rng.integers(0, len(vocab), rows_total)]
    vals = rng.integers(-1000, 1000, rows_total).astype(np.int64)
    agg = {}
    for v, x in zip(vals_s, vals):
        c, sm = agg.get(v, (0, 0))
        agg[v] = (c + 1, sm + int(x))
    exp = sorted(
        (np.int64(np.uint64(fnv1a64(k.encode()))), c, sm)
        for k, (c, sm) in agg.items())
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_strkey_worker, args=(r, 2, 29537, q))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        mk, mc, ms = q.get(timeout=300)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    np.testing.assert_array_equal(mk, [e[0] for e in exp])
    np.testing.assert_array_equal(mc, [e[1] for e in exp])
    np.testing.assert_array_equal(ms, [e[2] for e in exp])


def test_gloo_hash_agg_merge_equals_full():
    """Distributed merge of per-rank HASH-aggregate tables (arbitrary
    sparse i64 keys -> gather + merge by key) equals the
    single-node aggregation."""
    rows_total, seed = 90_000, 33
    rng = np.random.default_rng(seed)
    base = rng.integers(-(1 << 60), 1 << 60, 5000).astype(np.int64)
    keys = base[rng.integers(0, len(base), rows_total)]
    vals = rng.integers(-1000, 1000, rows_total).astype(np.int64)
    uk, inv = np.unique(keys, return_inverse=True)
    fc = np.bincount(inv, minlength=len(uk)).astype(np.int64)
    fs = np.zeros(len(uk), dtype=np.int64)
    np.add.at(fs, inv, vals)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_hash_worker, args=(r, 2, 29531, q))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        mk, mc, ms = q.get(timeout=300)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    np.testing.assert_array_equal(mk, uk)
    np.testing.assert_array_equal(mc, fc)
    np.testing.assert_array_equal(ms, fs)
