"""GPU parity tests (run on a real MI355X via gpurun / the round-end driver).

Every test compares the product GPU path (libsdb_gpu, HIP kernels) against
the CPU oracle on identical inputs. Bar (BASELINE.json north_star): doc ids,
hit sets and integer counts bit-exact; BM25 scores bit-exact in practice
(term-major fp32 order both sides; tolerance budget 1e-5 relative).
"""

import numpy as np
import pytest

import serenedb_amd as sa
from oracle import pyoracle as po

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    c = sa.GpuContext(0)
    yield c
    c.close()


def make_corpus(seed, doc_count, sels):
    postings = [sa.synth_postings(seed, doc_count, t, s)
                for t, s in enumerate(sels)]
    norms = sa.synth_norms(seed, doc_count)
    blob = sa.build_segment(doc_count, postings, norms)
    return blob, postings, norms


def check_parity(ctx, blob, term_idx, boosts, k, min_match=1,
                 global_stats=None, segs=None):
    if segs is None:
        segs = [ctx.load_segment(blob)]
        blobs = [blob]
    else:
        blobs = blob
    hits, total = ctx.execute_topk(segs, term_idx, boosts, k,
                                   min_match=min_match,
                                   global_stats=global_stats)
    ohits, ototal = po.execute_topk(blobs, term_idx, boosts, k,
                                    min_match=min_match,
                                    global_stats=global_stats)
    assert total == ototal, f"matches {total} != {ototal}"
    assert len(hits) == len(ohits)
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(hits["segment"], ohits["segment"])
    np.testing.assert_array_equal(
        hits["score"].view(np.uint32), ohits["score"].view(np.uint32),
        err_msg="fp32 scores must be bit-exact (term-major order)")
    return hits, total


def test_decode_term_parity(ctx):
    blob, postings, _ = make_corpus(42, 300_000, [0.05, 0.02, 0.5, 0.0007])
    seg = ctx.load_segment(blob)
    for t, (docs, freqs) in enumerate(postings):
        if len(docs) == 0:
            continue
        ddocs, dfreqs = ctx.decode_term(seg, t, len(docs))
        np.testing.assert_array_equal(ddocs, docs)
        np.testing.assert_array_equal(dfreqs, freqs)


def test_topk_or_parity(ctx):
    blob, _, _ = make_corpus(43, 1_000_000, [0.10, 0.05, 0.02, 0.01])
    for k in (10, 1000):
        check_parity(ctx, blob, [0, 1, 2, 3], [1.0] * 4, k)


def test_topk_and_minmatch_parity(ctx):
    blob, _, _ = make_corpus(44, 500_000, [0.2, 0.1, 0.05])
    check_parity(ctx, blob, [0, 1, 2], [1.0] * 3, 100, min_match=3)  # AND
    check_parity(ctx, blob, [0, 1, 2], [1.0] * 3, 100, min_match=2)


def test_topk_boosts_and_k1b(ctx):
    blob, _, _ = make_corpus(45, 400_000, [0.05, 0.02])
    seg = ctx.load_segment(blob)
    hits, total = ctx.execute_topk([seg], [0, 1], [2.0, 0.5], 50,
                                   k1=0.9, b=0.4)
    ohits, ototal = po.execute_topk([blob], [0, 1], [2.0, 0.5], 50,
                                    k1=0.9, b=0.4)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(hits["score"], ohits["score"])


def test_topk_edge_cases(ctx):
    # single posting, empty term, k > matches, dense term (bitset blocks),
    # tails of every family
    doc_count = 10_000
    rng = np.random.default_rng(5)
    dense = np.sort(rng.choice(np.arange(1, 2000, dtype=np.uint32), 1500,
                               replace=False))
    postings = [
        (np.array([doc_count], dtype=np.uint32),
         np.array([7], dtype=np.uint32)),            # last-doc posting
        (np.array([], dtype=np.uint32), np.array([], dtype=np.uint32)),
        (dense, rng.integers(1, 4, len(dense)).astype(np.uint32)),
        (np.arange(1, 40, dtype=np.uint32) * 250,
         np.full(39, 2, dtype=np.uint32)),           # all-same tails
    ]
    norms = sa.synth_norms(3, doc_count)
    blob = sa.build_segment(doc_count, postings, norms)
    check_parity(ctx, blob, [0, 1, 2, 3], [1.0] * 4, 100)
    check_parity(ctx, blob, [1], [1.0], 10)           # empty result
    check_parity(ctx, blob, [0, 2], [1.0, 1.0], 5000)  # k > matches


def test_multi_segment(ctx):
    b1, _, _ = make_corpus(46, 120_000, [0.05, 0.02])
    b2, _, _ = make_corpus(47, 80_000, [0.03, 0.04])
    s1 = ctx.load_segment(b1)
    s2 = ctx.load_segment(b2)
    check_parity(ctx, [b1, b2], [0, 1], [1.0, 1.0], 200, segs=[s1, s2])


def test_sharded_global_stats(ctx):
    """one GPU, shard segments with injected global stats == full corpus"""
    seed, doc_count = 48, 200_000
    sels = [0.05, 0.02, 0.01]
    full, postings, norms = make_corpus(seed, doc_count, sels)
    half = doc_count // 2
    shards = [sa.build_synth_segment(seed, 1, half, sels),
              sa.build_synth_segment(seed, half + 1, doc_count, sels)]
    dwt = [len(d) for d, _ in postings]
    gstats = (doc_count, int(norms[1:].sum()), dwt)
    fhits, ftotal = po.execute_topk([full], [0, 1, 2], [1.0] * 3, 150)
    cands = []
    totals = 0
    for base, blob in zip((0, half), shards):
        seg = ctx.load_segment(blob)
        h, t = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 150,
                                global_stats=gstats)
        totals += t
        for x in h:
            cands.append((float(x["score"]), int(x["doc"]) + base))
    cands.sort(key=lambda sd: (-sd[0], sd[1]))
    assert totals == ftotal
    assert [d for _, d in cands[:150]] == [int(d) for d in fhits["doc"]]


def test_scan_agg_parity(ctx):
    import ctypes as CT

    rows = 2_000_000
    ngroups = 1024
    rng = np.random.default_rng(44)
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = rng.integers(0, 1 << 20, rows).astype(np.int64)
    v2 = rng.normal(0, 1, rows).astype(np.float32)

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

    cols = (ColView * 3)(
        ColView(keys.ctypes.data_as(CT.c_void_p).value, rows, 0),
        ColView(v1.ctypes.data_as(CT.c_void_p).value, rows, 0),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
    tab = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(rows),
                                CT.byref(tab))
    assert rc == 0, rc
    # predicate: v1 < c at ~10% selectivity
    c = int((1 << 20) * 0.1)
    preds = (PredSpec * 1)(PredSpec(1, 1, c, 0, 0, 0))
    aggs = (AggSpec * 3)(AggSpec(0, 0), AggSpec(1, 1), AggSpec(2, 2))
    out = (AggResult * (ngroups * 3))()
    passed = CT.c_uint64(0)
    rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs, 3,
                              out, CT.byref(passed))
    assert rc == 0, rc
    ocnt, osi, osf, opassed = po.scan_agg(keys, v1, v2, ngroups, pred_op=1,
                                          lo=c)
    assert passed.value == opassed
    gcnt = np.array([out[g * 3 + 0].i64 for g in range(ngroups)])
    gsi = np.array([out[g * 3 + 1].i64 for g in range(ngroups)])
    gsf = np.array([out[g * 3 + 2].f64 for g in range(ngroups)])
    np.testing.assert_array_equal(gcnt, ocnt)
    np.testing.assert_array_equal(gsi, osi)
    np.testing.assert_allclose(gsf, osf, rtol=1e-7)
    lib.sdb_gpu_table_free(ctx._ctx, tab)


def test_smoke_entry():
    import __graft_entry__ as ge

    ge.smoke()


def test_hybrid_parity(ctx):
    """BM25 top-k AND column BETWEEN + bucket aggregates vs oracle
    (BASELINE configs[3] semantics)."""
    doc_count = 400_000
    sels = [0.1, 0.05, 0.02, 0.01]
    seed = 52
    blob, _, _ = make_corpus(seed, doc_count, sels)
    rng = np.random.default_rng(45)
    col = rng.integers(0, 1 << 31, doc_count + 1).astype(np.int64)
    span = 1 << 31
    flo, fhi = int(span * 0.4), int(span * 0.6) - 1
    nbuckets = 64
    seg = ctx.load_segment(blob)
    ctx.attach_column(seg, col)
    hits, total, bcnt, bsum = ctx.execute_topk_hybrid(
        [seg], [0, 1, 2, 3], [1.0] * 4, 1000, flo, fhi, nbuckets)
    ohits, ototal, obcnt, obsum = po.execute_topk_hybrid(
        blob, [0, 1, 2, 3], [1.0] * 4, 1000, col, flo, fhi, nbuckets)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(
        hits["score"].view(np.uint32), ohits["score"].view(np.uint32))
    np.testing.assert_array_equal(bcnt, obcnt)
    np.testing.assert_array_equal(bsum, obsum)


def test_bm15_parity(ctx):
    """BM15 (b=0): norm_const = k, no length normalization
    (bm25.cpp:296-299 stats branch)."""
    blob, _, _ = make_corpus(53, 200_000, [0.05, 0.02])
    seg = ctx.load_segment(blob)
    hits, total = ctx.execute_topk([seg], [0, 1], [1.0, 1.0], 100, b=0.0)
    ohits, ototal = po.execute_topk([blob], [0, 1], [1.0, 1.0], 100, b=0.0)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(
        hits["score"].view(np.uint32), ohits["score"].view(np.uint32))


def test_match_docs_parity(ctx):
    """streaming scan: GPU match emission + device column gather == oracle"""
    doc_count = 300_000
    blob, _, _ = make_corpus(62, doc_count, [0.05, 0.02, 0.01])
    col = np.random.default_rng(9).integers(0, 1 << 40, doc_count + 1
                                            ).astype(np.int64)
    seg = ctx.load_segment(blob)
    ctx.attach_column(seg, col)
    docs, vals, total = ctx.execute_match_docs(seg, [0, 1, 2], [1.0] * 3,
                                               doc_count, with_col=True)
    odocs, ovals, ototal = po.execute_match_docs(blob, [0, 1, 2], [1.0] * 3,
                                                 doc_count, col=col)
    assert total == ototal
    np.testing.assert_array_equal(docs, odocs)
    np.testing.assert_array_equal(vals, ovals)


def test_scan_agg_for_codec_parity(ctx):
    """FoR/bitpack columns + zonemap skips == oracle over the raw values.
    Keys clustered so zonemaps actually kill row groups under a BETWEEN."""
    import ctypes as CT

    rows = 3_000_000
    ngroups = 1024
    rng = np.random.default_rng(46)
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = np.sort(rng.integers(0, 1 << 20, rows)).astype(np.int64)  # clustered
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    keys_blob = sa.encode_col_i64(keys)
    v1_blob = sa.encode_col_i64(v1)

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

    kb = np.frombuffer(keys_blob, dtype=np.uint8)
    vb = np.frombuffer(v1_blob, dtype=np.uint8)
    cols = (ColView * 3)(
        ColView(kb.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(vb.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
    tab = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(rows),
                                CT.byref(tab))
    assert rc == 0, rc
    # clustered v1 + BETWEEN in the middle: most groups zonemap-dead
    lo_v = int((1 << 20) * 0.45)
    hi_v = int((1 << 20) * 0.55)
    preds = (PredSpec * 1)(PredSpec(1, 3, lo_v, hi_v, 0, 0))
    aggs = (AggSpec * 3)(AggSpec(0, 0), AggSpec(1, 1), AggSpec(2, 2))
    out = (AggResult * (ngroups * 3))()
    passed = CT.c_uint64(0)
    rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs, 3,
                              out, CT.byref(passed))
    assert rc == 0, rc
    ocnt, osi, osf, opassed = po.scan_agg(keys, v1, v2, ngroups, pred_op=3,
                                          lo=lo_v, hi=hi_v)
    assert passed.value == opassed
    gcnt = np.array([out[g * 3 + 0].i64 for g in range(ngroups)])
    gsi = np.array([out[g * 3 + 1].i64 for g in range(ngroups)])
    gsf = np.array([out[g * 3 + 2].f64 for g in range(ngroups)])
    np.testing.assert_array_equal(gcnt, ocnt)
    np.testing.assert_array_equal(gsi, osi)
    np.testing.assert_allclose(gsf, osf, rtol=1e-7)
    lib.sdb_gpu_table_free(ctx._ctx, tab)


def test_scan_agg_for_staged_edges(ctx):
    """Staged-FoR walker edge cases: odd row count with a 1-row tail group,
    a width-0 (constant) group, raw key + FoR predicate mix, a SUM over an
    unstaged FoR column (global col_read inside the staged kernel), pred on
    the key column (shared stage slot), and staged == unstaged
    (SDB_SCAN_NOSTAGE) on the same inputs."""
    import ctypes as CT
    import os

    rows = 65536 * 3 + 1  # 3 full groups + a 1-row tail group
    ngroups = 512
    rng = np.random.default_rng(47)
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = rng.integers(0, 1 << 20, rows).astype(np.int64)
    v1[:65536] = 5000  # constant group -> width 0, staging skipped
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    v3 = rng.integers(-(1 << 30), 1 << 30, rows).astype(np.int64)
    v1_blob = sa.encode_col_i64(v1)
    v3_blob = sa.encode_col_i64(v3)

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

    vb1 = np.frombuffer(v1_blob, dtype=np.uint8)
    vb3 = np.frombuffer(v3_blob, dtype=np.uint8)
    cols = (ColView * 4)(
        ColView(keys.ctypes.data_as(CT.c_void_p).value, rows, 0),  # raw key
        ColView(vb1.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1),
        ColView(vb3.ctypes.data_as(CT.c_void_p).value, rows, 2))
    tab = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols, 4, CT.c_uint64(rows),
                                CT.byref(tab))
    assert rc == 0, rc
    lo = 4000  # constant region (5000) passes
    preds = (PredSpec * 1)(PredSpec(1, 2, lo, 0, 0, 0))  # v1 >= lo
    aggs = (AggSpec * 3)(AggSpec(0, 0), AggSpec(3, 1), AggSpec(2, 2))
    out = (AggResult * (ngroups * 3))()
    passed = CT.c_uint64(0)

    def run():
        rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1, aggs,
                                  3, out, CT.byref(passed))
        assert rc == 0, rc
        return (np.array([out[g * 3 + 0].i64 for g in range(ngroups)]),
                np.array([out[g * 3 + 1].i64 for g in range(ngroups)]),
                np.array([out[g * 3 + 2].f64 for g in range(ngroups)]),
                passed.value)

    gcnt, gsi, gsf, gpassed = run()
    mask = v1 >= lo
    ecnt = np.bincount(keys[mask], minlength=ngroups)
    # f64 bincount is exact here: |v3| < 2^31 and |sums| < 2^53
    esi = np.bincount(keys[mask], weights=v3[mask].astype(np.float64),
                      minlength=ngroups).astype(np.int64)
    esf = np.bincount(keys[mask], weights=v2[mask].astype(np.float64),
                      minlength=ngroups)
    assert gpassed == int(mask.sum())
    np.testing.assert_array_equal(gcnt, ecnt)
    np.testing.assert_array_equal(gsi, esi)
    np.testing.assert_allclose(gsf, esf, rtol=1e-7, atol=1e-6)

    # unstaged walker on the same table must agree (ints bit-exact; the f64
    # sums only to atomic-order tolerance)
    os.environ["SDB_SCAN_NOSTAGE"] = "1"
    try:
        ucnt, usi, usf, upassed = run()
    finally:
        del os.environ["SDB_SCAN_NOSTAGE"]
    assert upassed == gpassed
    np.testing.assert_array_equal(ucnt, gcnt)
    np.testing.assert_array_equal(usi, gsi)
    np.testing.assert_allclose(usf, gsf, rtol=1e-7, atol=1e-6)
    lib.sdb_gpu_table_free(ctx._ctx, tab)

    # three predicates over FoR columns: preds 0/1 ride the stage slots,
    # pred 2 exercises the staged kernel's global col_read slow path
    cols3 = (ColView * 4)(
        ColView(keys.ctypes.data_as(CT.c_void_p).value, rows, 0),
        ColView(vb1.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1),
        ColView(vb3.ctypes.data_as(CT.c_void_p).value, rows, 2))
    tab3 = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols3, 4, CT.c_uint64(rows),
                                CT.byref(tab3))
    assert rc == 0, rc
    preds3 = (PredSpec * 3)(
        PredSpec(1, 2, lo, 0, 0, 0),            # v1 >= lo (staged)
        PredSpec(3, 2, -(1 << 29), 0, 0, 0),    # v3 >= -2^29 (staged)
        PredSpec(3, 1, 1 << 29, 0, 0, 0))       # v3 < 2^29 (global path)
    aggs3 = (AggSpec * 2)(AggSpec(0, 0), AggSpec(3, 1))
    out3 = (AggResult * (ngroups * 2))()
    rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab3, 0, ngroups, preds3, 3, aggs3,
                              2, out3, CT.byref(passed))
    assert rc == 0, rc
    m3 = (v1 >= lo) & (v3 >= -(1 << 29)) & (v3 < (1 << 29))
    assert passed.value == int(m3.sum())
    np.testing.assert_array_equal(
        np.array([out3[g * 2 + 0].i64 for g in range(ngroups)]),
        np.bincount(keys[m3], minlength=ngroups))
    np.testing.assert_array_equal(
        np.array([out3[g * 2 + 1].i64 for g in range(ngroups)]),
        np.bincount(keys[m3], weights=v3[m3].astype(np.float64),
                    minlength=ngroups).astype(np.int64))
    lib.sdb_gpu_table_free(ctx._ctx, tab3)

    # FoR key with the predicate on the key column itself: shared stage
    # slot (add_stage dedup) + agg_src=9 (key-as-value SUM)
    keys_blob = sa.encode_col_i64(keys)
    kb = np.frombuffer(keys_blob, dtype=np.uint8)
    cols2 = (ColView * 2)(
        ColView(kb.ctypes.data_as(CT.c_void_p).value, rows, 2),
        ColView(v2.ctypes.data_as(CT.c_void_p).value, rows, 1))
    tab2 = CT.c_void_p(0)
    rc = lib.sdb_gpu_table_load(ctx._ctx, cols2, 2, CT.c_uint64(rows),
                                CT.byref(tab2))
    assert rc == 0, rc
    preds2 = (PredSpec * 1)(PredSpec(0, 3, 100, 300, 0, 0))  # key BETWEEN
    aggs2 = (AggSpec * 2)(AggSpec(0, 0), AggSpec(0, 1))
    out2 = (AggResult * (ngroups * 2))()
    rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab2, 0, ngroups, preds2, 1, aggs2,
                              2, out2, CT.byref(passed))
    assert rc == 0, rc
    kmask = (keys >= 100) & (keys <= 300)
    ecnt2 = np.bincount(keys[kmask], minlength=ngroups)
    np.testing.assert_array_equal(
        np.array([out2[g * 2 + 0].i64 for g in range(ngroups)]), ecnt2)
    np.testing.assert_array_equal(
        np.array([out2[g * 2 + 1].i64 for g in range(ngroups)]),
        ecnt2 * np.arange(ngroups))
    assert passed.value == int(kmask.sum())
    lib.sdb_gpu_table_free(ctx._ctx, tab2)


def test_full_size_properties(ctx):
    """Size-independent properties at the full headline config (100M docs,
    the oracle is too slow to replay here): exact match count vs the
    independently-generated postings union; top-k ordering; threshold
    dominance (instruction ③: full-size property coverage)."""
    doc_count = 100_000_000
    sels = [0.10, 0.05, 0.02, 0.01]
    blob = sa.build_synth_segment(43, 1, doc_count, sels)
    seg = ctx.load_segment(blob)
    hits, total = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 1000)
    # exact union count from the deterministic corpus generator
    union = None
    for t, s in enumerate(sels):
        docs, _ = sa.synth_postings(43, doc_count, t, s)
        union = docs if union is None else np.union1d(union, docs)
    assert total == len(union)
    assert len(hits) == 1000
    s = hits["score"]
    assert np.all(s[:-1] >= s[1:])  # descending
    assert np.all(np.isin(hits["doc"], union))
    # determinism: a second run returns the identical result
    hits2, total2 = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 1000)
    assert total2 == total
    np.testing.assert_array_equal(hits["doc"], hits2["doc"])
    np.testing.assert_array_equal(hits["score"], hits2["score"])


def test_tfidf_parity(ctx):
    """TFIDF scorers (tfidf.cpp:60-76,148-151) through the same interface."""
    blob, _, _ = make_corpus(57, 300_000, [0.05, 0.02, 0.01])
    seg = ctx.load_segment(blob)
    for scorer in ("tfidf", "tfidf_norm"):
        hits, total = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 200,
                                       scorer=scorer)
        ohits, ototal = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 200,
                                        scorer=scorer)
        assert total == ototal
        np.testing.assert_array_equal(hits["doc"], ohits["doc"])
        np.testing.assert_array_equal(
            hits["score"].view(np.uint32), ohits["score"].view(np.uint32))


def test_wand_exactness(ctx):
    """WAND block-max pruning must return the IDENTICAL top-k (hits and
    scores) as the unpruned path — the reference's WAND is exact
    (formats_15_tests.cpp AssertWandPostings pins pruning correctness)."""
    blob, _, _ = make_corpus(58, 2_000_000, [0.10, 0.05, 0.02, 0.01])
    seg = ctx.load_segment(blob)
    for k in (10, 100, 1000):
        for scorer in ("bm25", "tfidf_norm"):
            base, _bt = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, k,
                                         scorer=scorer)
            wand, _wt = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, k,
                                         scorer=scorer, wand=True)
            np.testing.assert_array_equal(base["doc"], wand["doc"])
            np.testing.assert_array_equal(
                base["score"].view(np.uint32), wand["score"].view(np.uint32))


def test_eight_term_disjunction(ctx):
    sels = [0.08, 0.05, 0.04, 0.03, 0.02, 0.015, 0.01, 0.005]
    blob, _, _ = make_corpus(59, 400_000, sels)
    check_parity(ctx, blob, list(range(8)),
                 [1.0, 2.0, 0.5, 1.0, 3.0, 1.0, 0.25, 1.0], 300)
    check_parity(ctx, blob, list(range(8)), [1.0] * 8, 100, min_match=3)


def test_wide_plans(ctx):
    """20- and 32-term plans (the LDS desc cache shrinks per term count;
    blocks past the cache fall back to global desc reads)."""
    rng = np.random.default_rng(71)
    sels20 = [float(s) for s in rng.uniform(0.004, 0.05, 20)]
    blob, _, _ = make_corpus(72, 300_000, sels20)
    boosts = [float(b) for b in rng.uniform(0.25, 4.0, 20)]
    check_parity(ctx, blob, list(range(20)), boosts, 250)
    check_parity(ctx, blob, list(range(20)), [1.0] * 20, 100, min_match=5)

    sels32 = [float(s) for s in rng.uniform(0.003, 0.04, 32)]
    blob32, _, _ = make_corpus(73, 200_000, sels32)
    check_parity(ctx, blob32, list(range(32)), [1.0] * 32, 500)
    check_parity(ctx, blob32, list(range(32)), [1.0] * 32, 50, min_match=12)
    # > SDB_MAX_TERMS is rejected, not silently truncated
    import ctypes as CT
    seg = ctx.load_segment(blob32)
    hits = (sa.SdbScoreDoc * 10)()
    n = CT.c_uint32(0)
    tm = CT.c_uint64(0)
    plan = ctx._make_plan([0] * 33, [1.0] * 33, 1, 1.2, 0.75, None)
    rc = sa.gpu().sdb_gpu_execute_topk(
        ctx._ctx, (CT.c_void_p * 1)(CT.c_void_p(seg.value)), 1,
        CT.byref(plan), 10, hits, CT.byref(n), CT.byref(tm))
    assert rc == -1  # SDB_ERR_INVALID


def test_wand_dense_term_exactness(ctx):
    """WAND with a term denser than the desc cache depth (>32 blocks per
    24576-doc window at sel 0.25): the window bound must cover the unstaged
    descriptor tail, or pruning drops true hits."""
    blob, _, _ = make_corpus(74, 400_000, [0.25, 0.01])
    seg = ctx.load_segment(blob)
    for k in (10, 100):
        base, bt = ctx.execute_topk([seg], [0, 1], [1.0, 3.0], k)
        wand, _wt = ctx.execute_topk([seg], [0, 1], [1.0, 3.0], k, wand=True)
        assert bt >= _wt  # wand counts only visited matches
        np.testing.assert_array_equal(base["doc"], wand["doc"])
        np.testing.assert_array_equal(
            base["score"].view(np.uint32), wand["score"].view(np.uint32))


def test_hybrid_chain_parity(ctx):
    """Predicate-chain hybrid vs oracle: 2- and 3-pred chains over distinct
    column slots (BETWEEN + GE + LT), plus vacuous chain == single-pred."""
    doc_count = 300_000
    blob, _, _ = make_corpus(77, doc_count, [0.08, 0.04, 0.02])
    rng = np.random.default_rng(11)
    span = 1 << 30
    col0 = rng.integers(0, span, doc_count + 1).astype(np.int64)
    col1 = rng.integers(0, span, doc_count + 1).astype(np.int64)
    col2 = rng.integers(0, span, doc_count + 1).astype(np.int64)
    seg = ctx.load_segment(blob)
    ctx.attach_column(seg, col0, slot=0)
    ctx.attach_column(seg, col1, slot=1)
    ctx.attach_column(seg, col2, slot=2)
    flo, fhi = int(span * 0.1), int(span * 0.9) - 1
    g1 = int(span * 0.4)
    l2 = int(span * 0.7)
    nb = 32
    for preds, ops, los, his in (
        ([(0, 3, flo, fhi), (1, 2, g1, 0)], [3, 2], [flo, g1], [fhi, 0]),
        ([(0, 3, flo, fhi), (1, 2, g1, 0), (2, 1, l2, 0)],
         [3, 2, 1], [flo, g1, l2], [fhi, 0, 0]),
    ):
        hits, total, bcnt, bsum = ctx.execute_topk_hybrid_chain(
            [seg], [0, 1, 2], [1.0] * 3, 500, preds, nb)
        cols = [col0, col1, col2][:len(preds)]
        ohits, ototal, ocnt, osum = po.execute_topk_hybrid_chain(
            blob, [0, 1, 2], [1.0] * 3, 500, cols, ops, los, his, nb)
        assert total == ototal
        np.testing.assert_array_equal(hits["doc"], ohits["doc"])
        np.testing.assert_array_equal(
            hits["score"].view(np.uint32), ohits["score"].view(np.uint32))
        np.testing.assert_array_equal(bcnt, ocnt)
        np.testing.assert_array_equal(bsum, osum)
    # vacuous extra pred == single-pred hybrid (same GPU box, same inputs)
    h1, t1, c1, s1 = ctx.execute_topk_hybrid(
        [seg], [0, 1, 2], [1.0] * 3, 200, flo, fhi, nb)
    h2, t2, c2, s2 = ctx.execute_topk_hybrid_chain(
        [seg], [0, 1, 2], [1.0] * 3, 200, [(0, 3, flo, fhi), (1, 2, 0, 0)],
        nb)
    assert t1 == t2
    np.testing.assert_array_equal(h1["doc"], h2["doc"])
    np.testing.assert_array_equal(c1, c2)
    np.testing.assert_array_equal(s1, s2)
    # unattached slot / non-BETWEEN primary rejected
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        ctx.execute_topk_hybrid_chain([seg], [0], [1.0], 10,
                                      [(3, 3, 0, 10)], nb)
    with _pytest.raises(RuntimeError):
        ctx.execute_topk_hybrid_chain([seg], [0], [1.0], 10,
                                      [(0, 2, 0, 0)], nb)


def test_wand_pruning_fires_and_stays_exact(ctx):
    """Block-max pruning on the shape it exists for (skewed term
    frequencies, the BMW paper's motivating case): rare freq spikes set the
    k-th threshold; every block whose max_freq descriptor bound falls below
    it is skipped without decoding. Must skip real work (visited < total)
    and still return the exact top-k."""
    doc_count = 20_000_000
    docs = np.arange(1, doc_count + 1, 10, dtype=np.uint32)  # 2M postings
    freqs = np.ones(len(docs), dtype=np.uint32)
    freqs[::1009] = 200  # spike ~1 posting per 8 blocks
    norms = sa.synth_norms(78, doc_count)
    blob = sa.build_segment(doc_count, [(docs, freqs)], norms)
    seg = ctx.load_segment(blob)
    k = 10
    base, total = ctx.execute_topk([seg], [0], [1.0], k)
    wand, visited = ctx.execute_topk([seg], [0], [1.0], k, wand=True)
    np.testing.assert_array_equal(base["doc"], wand["doc"])
    np.testing.assert_array_equal(
        base["score"].view(np.uint32), wand["score"].view(np.uint32))
    # spike blocks are ~1/8 of all blocks; everything else prunes once the
    # threshold locks onto the spike scores
    assert visited < total // 2, (visited, total)


def test_filter_boost_parity(ctx):
    """Per-doc filter boost (HasFilterBoost variants): GPU == oracle
    bit-exact, incl. BM1's only nonzero form and WAND exactness under the
    scaled bounds; filter_boost without an attached column is rejected."""
    doc_count = 300_000
    blob, _, _ = make_corpus(80, doc_count, [0.08, 0.04, 0.02])
    rng = np.random.default_rng(14)
    fb = rng.uniform(0.5, 2.0, doc_count + 1).astype(np.float32)
    fb[0] = 0.0
    seg = ctx.load_segment(blob)
    import pytest as _pytest
    with _pytest.raises(RuntimeError):  # no boost column attached yet
        ctx.execute_topk([seg], [0, 1], [1.0] * 2, 10, filter_boost=True)
    ctx.attach_boost(seg, fb)
    for kwargs in ({}, {"min_match": 2}, {"k1": 0.0}, {"scorer": "tfidf"}):
        hits, total = ctx.execute_topk([seg], [0, 1, 2], [1.0, 2.0, 0.5],
                                       400, filter_boost=True, **kwargs)
        ohits, ototal = po.execute_topk([blob], [0, 1, 2], [1.0, 2.0, 0.5],
                                        400, filter_boost=fb, **kwargs)
        assert total == ototal
        np.testing.assert_array_equal(hits["doc"], ohits["doc"])
        np.testing.assert_array_equal(
            hits["score"].view(np.uint32), ohits["score"].view(np.uint32))
    # WAND stays exact with boosted bounds
    base, _t = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 50,
                                filter_boost=True)
    wnd, _v = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 50,
                               filter_boost=True, wand=True)
    np.testing.assert_array_equal(base["doc"], wnd["doc"])
    np.testing.assert_array_equal(
        base["score"].view(np.uint32), wnd["score"].view(np.uint32))


def test_bm1_parity(ctx):
    """BM1 (k1=0): empty top-k with exact match counting on both sides."""
    blob, _, _ = make_corpus(75, 200_000, [0.05, 0.02])
    seg = ctx.load_segment(blob)
    hits, total = ctx.execute_topk([seg], [0, 1], [1.0, 1.0], 100, k1=0.0)
    ohits, ototal = po.execute_topk([blob], [0, 1], [1.0, 1.0], 100, k1=0.0)
    assert len(hits) == 0 and len(ohits) == 0
    assert total == ototal > 0


def test_match_docs_multi_segment(ctx):
    """streaming scan across segments: per-segment loop (the reference's
    worker claims segments one at a time), (segment, doc) ascending, column
    gather per segment, cap cuts emission but not total_matches."""
    seed = 76
    sels = [0.05, 0.02]
    n1, n2 = 150_000, 120_000
    b1 = sa.build_synth_segment(seed, 1, n1, sels)
    b2 = sa.build_synth_segment(seed, n1 + 1, n1 + n2, sels)
    rng = np.random.default_rng(8)
    col1 = rng.integers(0, 1 << 40, n1 + 1).astype(np.int64)
    col2 = rng.integers(0, 1 << 40, n2 + 1).astype(np.int64)
    s1 = ctx.load_segment(b1)
    s2 = ctx.load_segment(b2)
    ctx.attach_column(s1, col1)
    ctx.attach_column(s2, col2)
    segs_out, docs, vals, total = ctx.execute_match_docs_multi(
        [s1, s2], [0, 1], [1.0, 1.0], n1 + n2, with_col=True)
    exp = []
    tot = 0
    for si, (blob, col) in enumerate(((b1, col1), (b2, col2))):
        od, ov, ot = po.execute_match_docs(blob, [0, 1], [1.0, 1.0],
                                           n1 + n2, col=col)
        tot += ot
        exp.extend((si, int(d), int(v)) for d, v in zip(od, ov))
    assert total == tot
    got = list(zip(segs_out.tolist(), docs.tolist(), vals.tolist()))
    assert got == exp
    # cap smaller than the first segment's matches: emission truncated,
    # total still covers every segment
    cap = len([e for e in exp if e[0] == 0]) // 2
    segs_c, docs_c, _, total_c = ctx.execute_match_docs_multi(
        [s1, s2], [0, 1], [1.0, 1.0], cap, with_col=False)
    assert total_c == tot
    assert len(docs_c) == cap
    assert got[:cap] == list(zip(segs_c.tolist(), docs_c.tolist(),
                                 [e[2] for e in exp[:cap]]))


def test_hybrid_multi_segment(ctx):
    """hybrid across two resident segments: bucket aggregates accumulate
    and the merged top-k equals per-segment oracle runs merged with global
    stats (the multi-segment PreparePhase semantics)."""
    seed = 63
    sels = [0.08, 0.03]
    n1, n2 = 150_000, 100_000
    b1 = sa.build_synth_segment(seed, 1, n1, sels)
    b2 = sa.build_synth_segment(seed, n1 + 1, n1 + n2, sels)
    rng = np.random.default_rng(12)
    span = 1 << 31
    col1 = rng.integers(0, span, n1 + 1).astype(np.int64)
    col2 = rng.integers(0, span, n2 + 1).astype(np.int64)
    flo, fhi = int(span * 0.3), int(span * 0.7) - 1
    nb = 32
    s1 = ctx.load_segment(b1)
    s2 = ctx.load_segment(b2)
    ctx.attach_column(s1, col1)
    ctx.attach_column(s2, col2)
    hits, total, bcnt, bsum = ctx.execute_topk_hybrid(
        [s1, s2], [0, 1], [1.0, 1.0], 200, flo, fhi, nb)
    # oracle: per-segment with merged global stats, buckets summed
    d0 = sa.synth_postings(seed, n1 + n2, 0, sels[0])[0]
    d1 = sa.synth_postings(seed, n1 + n2, 1, sels[1])[0]
    norms = sa.synth_norms(seed, n1 + n2)
    gstats = (n1 + n2, int(norms[1:].sum()), [len(d0), len(d1)])
    tot = 0
    cands = []
    cnt = np.zeros(nb, dtype=np.int64)
    sm = np.zeros(nb, dtype=np.int64)
    for si, (blob, col) in enumerate(((b1, col1), (b2, col2))):
        h, t, bc, bs = po.execute_topk_hybrid(
            blob, [0, 1], [1.0, 1.0], 200, col, flo, fhi, nb,
            global_stats=gstats)
        tot += t
        cnt += bc
        sm += bs
        for x in h:
            cands.append((float(x["score"]), si, int(x["doc"])))
    cands.sort(key=lambda v: (-v[0], v[1], v[2]))
    assert total == tot
    np.testing.assert_array_equal(bcnt, cnt)
    np.testing.assert_array_equal(bsum, sm)
    got = [(float(h["score"]), int(h["segment"]), int(h["doc"]))
           for h in hits]
    assert got == cands[:len(got)]


def test_count_fast(ctx):
    """CountFast (docs-only decode) == full-path total_matches, OR and
    min-match variants."""
    blob, _, _ = make_corpus(64, 800_000, [0.08, 0.04, 0.01])
    seg = ctx.load_segment(blob)
    for mm in (1, 2, 3):
        _, full = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 10,
                                   min_match=mm)
        fast = ctx.execute_count([seg], [0, 1, 2], [1.0] * 3, min_match=mm)
        assert fast == full, (mm, fast, full)


def test_hybrid_and_minmatch(ctx):
    """hybrid with conjunction semantics (min_match = nterms)"""
    blob, _, _ = make_corpus(66, 300_000, [0.15, 0.1])
    rng = np.random.default_rng(8)
    col = rng.integers(0, 1 << 30, 300_001).astype(np.int64)
    flo, fhi = 0, (1 << 29) - 1  # ~50%
    seg = ctx.load_segment(blob)
    ctx.attach_column(seg, col)
    hits, total, bcnt, bsum = ctx.execute_topk_hybrid(
        [seg], [0, 1], [1.0, 1.0], 100, flo, fhi, 16, min_match=2)
    ohits, ototal, obc, obs = po.execute_topk_hybrid(
        blob, [0, 1], [1.0, 1.0], 100, col, flo, fhi, 16, min_match=2)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(bcnt, obc)
    np.testing.assert_array_equal(bsum, obs)


def test_duplicate_terms_rejected(ctx):
    blob, _, _ = make_corpus(67, 10_000, [0.1, 0.05])
    seg = ctx.load_segment(blob)
    import pytest as _pt

    with _pt.raises(RuntimeError):
        ctx.execute_topk([seg], [0, 0], [1.0, 1.0], 10)


def test_threshold_ties_at_kth(ctx):
    """Directed bin-threshold exactness (round-1 VERDICT weak #2): plant
    large tie classes AT the k-th boundary so any off-by-one between the
    histogram bucketing and the append/final filter drops (or duplicates)
    members of the boundary tie class. The bin-space threshold — the same
    (u32)(s*inv_smax) expression counted, compared and host-filtered —
    must keep the hit set identical to the oracle's."""
    doc_count = 200_000
    docs = np.arange(1, doc_count + 1, dtype=np.uint32)

    # (a) ALL scores identical: every matched doc shares one bin; the k-th
    # boundary splits a single tie class (threshold bin == that bin)
    freqs = np.ones(doc_count, dtype=np.uint32)
    norms = np.full(doc_count + 1, 7, dtype=np.uint32)
    blob = sa.build_segment(doc_count, [(docs, freqs)], norms)
    check_parity(ctx, blob, [0], [1.0], 1000)

    # (b) two tie classes with the boundary inside the lower class: 3000
    # docs at freq=4 (higher score), the rest at freq=1, k=3500
    freqs2 = np.ones(doc_count, dtype=np.uint32)
    freqs2[::67] = 4  # ~2985 high-score docs interleaved across windows
    blob2 = sa.build_segment(doc_count, [(docs, freqs2)], norms)
    check_parity(ctx, blob2, [0], [1.0], 3500)

    # (c) k exceeds the number of matches: no threshold ever derives;
    # every match must come back
    check_parity(ctx, blob2, [0], [1.0], 250_000)

    # (d) TFIDF score exactly at a power-of-two bin edge: freq in {1,4}
    # puts s = num (sqrt(1)) at smax/2 = bin 128's edge
    hits, _ = ctx.execute_topk([ctx.load_segment(blob2)], [0], [1.0], 3500,
                               scorer="tfidf")
    ohits, _ = po.execute_topk([blob2], [0], [1.0], 3500, scorer="tfidf")
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(hits["score"].view(np.uint32),
                                  ohits["score"].view(np.uint32))


def test_bad_segment_rejected(ctx):
    """Corrupted/truncated segment blobs fail loudly at load:
    no section may extend past the blob and no term span past its
    section."""
    blob, _, _ = make_corpus(77, 50_000, [0.05])
    b = bytearray(blob)

    # header fields (sdb_format.h SdbSegHeader layout)
    def poke(off, val):
        bb = bytearray(blob)
        bb[off:off + 8] = int(val).to_bytes(8, "little")
        return bytes(bb)

    import pytest as _pytest
    # off_desc -> past the blob (offset 40 = off_terms, 48 = off_desc)
    for field_off in (40, 48, 56, 64):
        with _pytest.raises(RuntimeError):
            ctx.load_segment(poke(field_off, len(blob) + 4096))
    # total_blocks (offset 32) huge: desc section would overrun
    with _pytest.raises(RuntimeError):
        ctx.load_segment(poke(32, (1 << 62)))
    # truncated blob (cut inside the payload)
    with _pytest.raises(RuntimeError):
        ctx.load_segment(bytes(b[: len(b) // 2]))
    # term entry pointing past the desc section: term 0 desc_end huge
    hdr_off_terms = int.from_bytes(blob[40:48], "little")
    bb = bytearray(blob)
    bb[hdr_off_terms + 8:hdr_off_terms + 16] = int(1 << 61).to_bytes(
        8, "little")
    with _pytest.raises(RuntimeError):
        ctx.load_segment(bytes(bb))
    # pristine blob still loads
    ctx.load_segment(blob)


def test_negative_boost_rejected(ctx):
    """Negative/NaN boosts would break every non-negative-score assumption
    (histogram bins, bin threshold, WAND bounds) — rejected up front
    (ADVICE r1)."""
    import pytest as _pytest
    blob, _, _ = make_corpus(78, 20_000, [0.05, 0.02])
    seg = ctx.load_segment(blob)
    with _pytest.raises(RuntimeError):
        ctx.execute_topk([seg], [0, 1], [1.0, -0.5], 10)
    with _pytest.raises(RuntimeError):
        ctx.execute_topk([seg], [0], [float("nan")], 10)
    fb = np.ones(20_001, dtype=np.float32)
    fb[777] = -1.0
    with _pytest.raises(RuntimeError):
        ctx.attach_boost(seg, fb)


def test_dense_group_key_range_rejected(ctx):
    """scan_agg's perfect-hash kernel requires keys in [0, ngroups); an
    out-of-range key would scribble past the LDS accumulators. Both raw
    (load-time GPU min/max) and FoR (zonemaps) columns are validated
    (ADVICE r1 / VERDICT weak #4)."""
    import pytest as _pytest
    rows = 100_000
    rng = np.random.default_rng(5)
    keys = rng.integers(0, 64, rows).astype(np.int64)
    keys[12345] = 64  # == ngroups: out of range
    vals = rng.integers(0, 1000, rows).astype(np.int64)
    for enc in ("raw", "for"):
        tab = ctx.load_table_i64(keys, vals, codec=enc)
        with _pytest.raises(RuntimeError):
            ctx.scan_agg_count_sum(tab, ngroups=64)
        # with room for the stray key the same table aggregates fine
        out = ctx.scan_agg_count_sum(tab, ngroups=65)
        assert out is not None
        ctx.free_table(tab)


def _np_hash_agg(keys, v1, v2, sel):
    """independent numpy expected result for the hash aggregate"""
    ks = keys[sel]
    uk, inv = np.unique(ks, return_inverse=True)
    cnt = np.bincount(inv, minlength=len(uk))
    s1 = np.bincount(inv, weights=v1[sel].astype(np.float64),
                     minlength=len(uk)).astype(np.int64)
    # exact integer sums: use object-free path via add.at on int64
    s1 = np.zeros(len(uk), dtype=np.int64)
    np.add.at(s1, inv, v1[sel])
    s2 = np.zeros(len(uk), dtype=np.float64)
    np.add.at(s2, inv, v2[sel].astype(np.float64))
    return uk, cnt, s1, s2


def test_hash_agg_sparse_keys(ctx):
    """General hash group-by (north_star's LDS-staged open-addressed
    buckets; VERDICT #1 missing item): random sparse 64-bit keys, ~100k
    distinct groups, predicate pushdown — vs an independent numpy
    aggregation. Raw and FoR key codecs."""
    rows = 3_000_000
    rng = np.random.default_rng(99)
    # raw: full-range 64-bit keys; for: sparse keys within the FoR codec's
    # 32-bit group-width envelope (the codec is width<=32 by design)
    base_raw = rng.integers(-(1 << 62), 1 << 62, 100_000).astype(np.int64)
    base_for = rng.integers(-(1 << 30), 1 << 30, 100_000).astype(np.int64)
    v1 = rng.integers(-(1 << 30), 1 << 30, rows).astype(np.int64)
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    c = int((1 << 30) * -0.8)
    sel = v1 >= c

    for enc, base in (("raw", base_raw), ("for", base_for)):
        keys = base[rng.integers(0, len(base), rows)]
        keys[::997] = -1  # the sentinel-adjacent key must work too
        uk, cnt, s1, s2 = _np_hash_agg(keys, v1, v2, sel)
        tab = ctx.load_table([keys, v1, v2], [enc, enc, "raw"])
        gkeys, i64, f64, passed = ctx.scan_agg_hash(
            tab, 0, 200_000, [(1, 2, c, 0)], [(0, 0), (1, 1), (2, 2)])
        assert passed == int(sel.sum())
        np.testing.assert_array_equal(gkeys, uk)
        np.testing.assert_array_equal(i64[:, 0], cnt)
        np.testing.assert_array_equal(i64[:, 1], s1)
        np.testing.assert_allclose(f64[:, 2], s2, rtol=1e-7)
        # distinct keys above max_groups -> loud SDB_ERR_OOM
        import pytest as _pytest
        with _pytest.raises(RuntimeError):
            ctx.scan_agg_hash(tab, 0, 1000, [], [(0, 0)])
        ctx.free_table(tab)


def test_scan_eq_and_f32_predicates(ctx):
    """EQ + f32 predicate columns (VERDICT #7 / table_filter_iterator
    typed compares) across the dense and hash aggregate kernels."""
    rows = 1_000_000
    ngroups = 256
    rng = np.random.default_rng(31)
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = rng.integers(0, 1000, rows).astype(np.int64)
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    v2[::101] = np.nan  # NaN fails every compare

    tab = ctx.load_table([keys, v1, v2])
    # EQ on i64
    sel = v1 == 123
    i64, f64, passed = ctx.scan_agg(tab, 0, ngroups,
                                    [(1, 4, 123, 0)], [(0, 0), (1, 1)])
    assert passed == int(sel.sum())
    exp_cnt = np.bincount(keys[sel], minlength=ngroups)
    np.testing.assert_array_equal(i64[:, 0], exp_cnt)
    # f32 BETWEEN (NaN rows excluded)
    fsel = (v2 >= np.float32(-0.5)) & (v2 <= np.float32(0.5))
    i64b, f64b, passedb = ctx.scan_agg(
        tab, 0, ngroups, [(2, 3, -0.5, 0.5)], [(0, 0), (2, 2)])
    assert passedb == int(fsel.sum())
    np.testing.assert_array_equal(
        i64b[:, 0], np.bincount(keys[fsel], minlength=ngroups))
    s2 = np.zeros(ngroups, dtype=np.float64)
    np.add.at(s2, keys[fsel], v2[fsel].astype(np.float64))
    np.testing.assert_allclose(f64b[:, 1], s2, rtol=1e-7)
    # same through the hash kernel
    gk, hi64, hf64, hp = ctx.scan_agg_hash(
        tab, 0, ngroups + 8, [(2, 3, -0.5, 0.5)], [(0, 0)])
    assert hp == passedb
    live = np.nonzero(np.bincount(keys[fsel], minlength=ngroups))[0]
    np.testing.assert_array_equal(gk, live)
    ctx.free_table(tab)


def test_livemask_parity(ctx):
    """Deleted-doc live masks (round-1 missing #2; seg.mask(it) at
    duckdb_search_full_scan.cpp:1898, Masked count :2475): with a mask
    attached, hits, total_matches, counts, WAND results and streaming
    emission see LIVE docs only — vs the oracle with the same mask."""
    doc_count = 500_000
    blob, postings, _ = make_corpus(55, doc_count, [0.10, 0.05, 0.02])
    rng = np.random.default_rng(56)
    # ~25% deleted
    nwords = (doc_count + 64) // 64
    mask = rng.integers(0, 1 << 64, nwords, dtype=np.uint64)
    mask |= rng.integers(0, 1 << 64, nwords, dtype=np.uint64)
    seg = ctx.load_segment(blob)
    ctx.attach_livemask(seg, mask)

    def live(d):
        return (int(mask[d >> 6]) >> (d & 63)) & 1

    for mm in (1, 2, 3):  # lean sweep path (mm=1) and general path
        hits, total = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 500,
                                       min_match=mm)
        ohits, ototal = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 500,
                                        min_match=mm, live_mask=mask)
        assert total == ototal
        np.testing.assert_array_equal(hits["doc"], ohits["doc"])
        np.testing.assert_array_equal(hits["score"].view(np.uint32),
                                      ohits["score"].view(np.uint32))
        assert all(live(int(d)) for d in hits["doc"])

    # masked CountFast == oracle masked count
    cnt = ctx.execute_count([seg], [0, 1], [1.0, 1.0])
    _, ocnt = po.execute_topk([blob], [0, 1], [1.0, 1.0], 1,
                              live_mask=mask)
    assert cnt == ocnt

    # WAND stays exact under the mask
    h1, t1 = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 10, wand=True)
    h0, _ = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 10,
                            live_mask=mask)
    np.testing.assert_array_equal(h1["doc"], h0["doc"])
    np.testing.assert_array_equal(h1["score"].view(np.uint32),
                                  h0["score"].view(np.uint32))

    # streaming emission: only live docs come out
    docs, _, tot = ctx.execute_match_docs(seg, [0], [1.0], doc_count)
    exp = postings[0][0][[bool(live(int(d))) for d in postings[0][0]]]
    np.testing.assert_array_equal(docs, exp)
    assert tot == len(exp)

    # detach: everything matches the unmasked oracle again
    ctx.attach_livemask(seg, None)
    hits, total = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 100)
    ohits, ototal = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 100)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])


def test_batch_pipelined_equals_single(ctx):
    """sdb_gpu_execute_topk_batch (pipelined QPS shape) must return, for
    EVERY query in the batch, exactly what the single-query entry
    returns — nothing is cached or shared between queries beyond the
    resident segment."""
    blob, _, _ = make_corpus(61, 400_000, [0.1, 0.05, 0.02, 0.01])
    seg = ctx.load_segment(blob)
    single, stotal = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 777)
    batch, btotals = ctx.execute_topk_batch([seg], [0, 1, 2, 3], [1.0] * 4,
                                            777, 5, all_hits=True)
    assert btotals == [stotal] * 5
    for q in range(5):
        np.testing.assert_array_equal(batch[q]["doc"], single["doc"])
        np.testing.assert_array_equal(batch[q]["score"].view(np.uint32),
                                      single["score"].view(np.uint32))
    # min_match / AND plans route through the general kernel in batch too
    s2, t2 = ctx.execute_topk([seg], [0, 1], [1.0] * 2, 50, min_match=2)
    b2, bt2 = ctx.execute_topk_batch([seg], [0, 1], [1.0] * 2, 50, 3,
                                     min_match=2, all_hits=True)
    assert bt2 == [t2] * 3
    np.testing.assert_array_equal(b2[0]["doc"], s2["doc"])
    np.testing.assert_array_equal(b2[2]["doc"], s2["doc"])
    # MULTI-SEGMENT batch: term-slot rotation walks q*nsegs+sg — two
    # segments per query must still match the single-query entry
    blob2, _, _ = make_corpus(63, 300_000, [0.1, 0.05, 0.02, 0.01])
    seg2 = ctx.load_segment(blob2)
    sm, tm = ctx.execute_topk([seg, seg2], [0, 1, 2, 3], [1.0] * 4, 500)
    bm, btm = ctx.execute_topk_batch([seg, seg2], [0, 1, 2, 3], [1.0] * 4,
                                     500, 4, all_hits=True)
    assert btm == [tm] * 4
    for q in range(4):
        np.testing.assert_array_equal(bm[q]["doc"], sm["doc"])
        np.testing.assert_array_equal(bm[q]["segment"], sm["segment"])
        np.testing.assert_array_equal(bm[q]["score"].view(np.uint32),
                                      sm["score"].view(np.uint32))

    # WAND plans through the batch (plan.wand propagates per segment)
    sw, tw = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 10,
                              wand=True)
    bw, btw = ctx.execute_topk_batch([seg], [0, 1, 2, 3], [1.0] * 4, 10,
                                     3, wand=True, all_hits=True)
    assert btw == [tw] * 3
    for q in range(3):
        np.testing.assert_array_equal(bw[q]["doc"], sw["doc"])
        np.testing.assert_array_equal(bw[q]["score"].view(np.uint32),
                                      sw["score"].view(np.uint32))

    # hybrid batch: per-query hits AND bucket planes equal the single
    # hybrid entry (bucket state double-buffered by query parity)
    rng = np.random.default_rng(62)
    col = rng.integers(0, 1 << 20, 400_000 + 1).astype(np.int64)
    ctx.attach_column(seg, col)
    flo, fhi = 1000, (1 << 19)
    sh, st, sc, ss = ctx.execute_topk_hybrid([seg], [0, 1, 2, 3],
                                             [1.0] * 4, 777, flo, fhi, 64)
    bh, bt, bc, bs = ctx.execute_topk_hybrid_batch(
        [seg], [0, 1, 2, 3], [1.0] * 4, 777, flo, fhi, 64, 5,
        all_hits=True)
    assert bt == [st] * 5
    for q in range(5):
        np.testing.assert_array_equal(bh[q]["doc"], sh["doc"])
        np.testing.assert_array_equal(bh[q]["score"].view(np.uint32),
                                      sh["score"].view(np.uint32))
        np.testing.assert_array_equal(bc[q], sc)
        np.testing.assert_array_equal(bs[q], ss)


def test_validity_null_semantics(ctx):
    """Column validity (null) masks — SQL three-valued logic as the
    reference's pushed filters implement it (tests/fuzz/
    null_semantics_fuzz.py; table_filter_iterator.hpp NullCheckKind):
    comparisons drop NULL rows, IS [NOT] NULL evaluates the validity
    plane alone, SUM skips NULL values while COUNT(*) counts the row —
    vs an independent numpy evaluation, on dense, FoR and hash paths."""
    rows = 800_000
    ngroups = 128
    rng = np.random.default_rng(71)
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = rng.integers(0, 1000, rows).astype(np.int64)
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    valid1 = rng.random(rows) > 0.3   # 30% NULLs in v1
    valid2 = rng.random(rows) > 0.2   # 20% NULLs in v2

    def words(v):
        w = np.zeros((rows + 63) // 64, dtype=np.uint64)
        idx = np.nonzero(v)[0]
        np.bitwise_or.at(w, idx // 64,
                         (np.uint64(1) << (idx % 64).astype(np.uint64)))
        return w

    for enc in ("raw", "for"):
        tab = ctx.load_table([keys, v1, v2], [enc, enc, "raw"])
        ctx.attach_validity(tab, 1, words(valid1))
        ctx.attach_validity(tab, 2, words(valid2))

        # (a) comparison drops NULLs: v1 < 500
        sel = valid1 & (v1 < 500)
        i64, f64, passed = ctx.scan_agg(tab, 0, ngroups, [(1, 1, 500, 0)],
                                        [(0, 0), (1, 1), (2, 2)])
        assert passed == int(sel.sum())
        np.testing.assert_array_equal(
            i64[:, 0], np.bincount(keys[sel], minlength=ngroups))
        # SUM(v1) over selected rows (all valid there); SUM(v2) skips
        # v2-NULLs among the selected rows
        s1 = np.zeros(ngroups, dtype=np.int64)
        np.add.at(s1, keys[sel], v1[sel])
        np.testing.assert_array_equal(i64[:, 1], s1)
        sel2 = sel & valid2
        s2 = np.zeros(ngroups, dtype=np.float64)
        np.add.at(s2, keys[sel2], v2[sel2].astype(np.float64))
        np.testing.assert_allclose(f64[:, 2], s2, rtol=1e-7)

        # (b) IS NULL / IS NOT NULL on the validity plane alone
        i64b, _, pb = ctx.scan_agg(tab, 0, ngroups, [(1, 5, 0, 0)],
                                   [(0, 0)])
        assert pb == int((~valid1).sum())
        np.testing.assert_array_equal(
            i64b[:, 0], np.bincount(keys[~valid1], minlength=ngroups))
        _, _, pnn = ctx.scan_agg(tab, 0, ngroups, [(1, 6, 0, 0)], [(0, 0)])
        assert pnn == int(valid1.sum())
        # ISNULL on a column WITHOUT validity selects nothing
        _, _, p0 = ctx.scan_agg(tab, 0, ngroups, [(0, 5, 0, 0)], [(0, 0)])
        assert p0 == 0

        # (c) f32 predicate drops NULLs
        fsel = valid2 & (v2 >= np.float32(0.0))
        _, _, pf = ctx.scan_agg(tab, 0, ngroups, [(2, 2, 0.0, 0.0)],
                                [(0, 0)])
        assert pf == int(fsel.sum())

        # (d) hash path honors the same semantics
        gk, hi64, _, hp = ctx.scan_agg_hash(
            tab, 0, ngroups + 8, [(1, 5, 0, 0)], [(0, 0)])
        assert hp == pb
        live = np.nonzero(np.bincount(keys[~valid1],
                                      minlength=ngroups))[0]
        np.testing.assert_array_equal(gk, live)

        # (e) validity on the group key is rejected
        import pytest as _pytest
        ctx.attach_validity(tab, 0, words(valid1))
        with _pytest.raises(RuntimeError):
            ctx.scan_agg(tab, 0, ngroups, [], [(0, 0)])
        ctx.attach_validity(tab, 0, None)  # detach restores
        ctx.scan_agg(tab, 0, ngroups, [], [(0, 0)])
        ctx.free_table(tab)


def test_wave_kernel_path_parity(ctx):
    """The experimental barrier-free per-wave kernel (SDB_TOPK_PATH=wave)
    keeps full parity even though it is no longer the default path."""
    import os
    blob, _, _ = make_corpus(81, 600_000, [0.1, 0.05, 0.02, 0.01])
    seg = ctx.load_segment(blob)
    base, btotal = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 500)
    os.environ["SDB_TOPK_PATH"] = "wave"
    try:
        w, wtotal = ctx.execute_topk([seg], [0, 1, 2, 3], [1.0] * 4, 500)
    finally:
        os.environ.pop("SDB_TOPK_PATH", None)
    assert wtotal == btotal
    np.testing.assert_array_equal(w["doc"], base["doc"])
    np.testing.assert_array_equal(w["score"].view(np.uint32),
                                  base["score"].view(np.uint32))


def test_multi_context_isolation(ctx):
    """Two contexts on one device run interleaved queries without state
    leakage (contexts are single-threaded but independent: own streams,
    workspaces, thresholds — include/sdb_gpu.h contract)."""
    import serenedb_amd as sa

    blob, _, _ = make_corpus(71, 200_000, [0.1, 0.05])
    blob2, _, _ = make_corpus(72, 150_000, [0.2, 0.02])
    ctx2 = sa.GpuContext(0)
    seg1 = ctx.load_segment(blob)
    seg2 = ctx2.load_segment(blob2)
    r1, t1 = ctx.execute_topk([seg1], [0, 1], [1.0] * 2, 100)
    r2, t2 = ctx2.execute_topk([seg2], [0, 1], [1.0] * 2, 100)
    # interleave: each context re-runs its own query and must reproduce
    for _ in range(3):
        a1, b1 = ctx.execute_topk([seg1], [0, 1], [1.0] * 2, 100)
        a2, b2 = ctx2.execute_topk([seg2], [0, 1], [1.0] * 2, 100)
        assert b1 == t1 and b2 == t2
        np.testing.assert_array_equal(a1["doc"], r1["doc"])
        np.testing.assert_array_equal(a2["doc"], r2["doc"])
        np.testing.assert_array_equal(a1["score"].view(np.uint32),
                                      r1["score"].view(np.uint32))
        np.testing.assert_array_equal(a2["score"].view(np.uint32),
                                      r2["score"].view(np.uint32))


def test_table_load_free_cycles(ctx):
    """repeated load/scan/free cycles return stable results (no workspace
    reuse corruption, no handle confusion)"""
    rng = np.random.default_rng(73)
    rows = 500_000
    keys = rng.integers(0, 32, rows).astype(np.int64)
    v1 = rng.integers(0, 1000, rows).astype(np.int64)
    ref = None
    for _ in range(4):
        tab = ctx.load_table([keys, v1])
        i64, _, passed = ctx.scan_agg(tab, 0, 32, [], [(0, 0), (1, 1)])
        if ref is None:
            ref = (i64.copy(), passed)
        else:
            np.testing.assert_array_equal(i64, ref[0])
            assert passed == ref[1]
        ctx.free_table(tab)
