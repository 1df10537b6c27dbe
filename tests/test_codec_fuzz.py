"""Property-based codec fuzzing (hypothesis): the block codec is the parity
core (FormatTraits128, format_block_128.hpp:51-379 write / :446-636 read),
so the host encoder and the oracle restatement are cross-checked on
adversarial shapes the hand-written cases might miss: all-same runs that
break mid-block, gap spikes that force byte-width jumps inside streamvbyte,
dense regions that flip to bitset, maximum 32-bit doc ids, freq
distributions straddling every family boundary.

Both directions: host encode -> oracle decode and oracle encode -> host
decode must reproduce the input exactly (byte streams may differ only if an
encoder picks a different legal family — they don't: encoder selection is
deterministic and mirrored, which test_cross_encoder_bytes pins)."""

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import serenedb_amd as sa
from oracle import pyoracle as po


def docs_strategy():
    """ascending doc-id blocks (<=128) built from segments of adversarial
    delta patterns: all-same runs, unit steps (bitset bait), huge gaps."""
    seg = st.one_of(
        st.tuples(st.integers(1, 64), st.integers(1, 4)),          # run of same small delta
        st.tuples(st.integers(1, 32), st.just(1)),                 # dense run
        st.tuples(st.integers(1, 4), st.integers(1, 1 << 24)),     # gap spikes
        st.tuples(st.integers(1, 16), st.integers(1, 1 << 14)),
    )
    return st.lists(seg, min_size=1, max_size=8).map(_segments_to_docs)


def _segments_to_docs(segs):
    deltas = []
    for n, d in segs:
        deltas.extend([d] * n)
    deltas = deltas[:128]
    docs = np.cumsum(np.array(deltas, dtype=np.uint64))
    docs = docs[docs <= 0xFFFFFFFE]
    return docs.astype(np.uint32)


@st.composite
def freqs_strategy(draw):
    n = draw(st.integers(1, 128))
    kind = draw(st.integers(0, 3))
    if kind == 0:  # all-same (1..4-byte widths)
        v = draw(st.sampled_from([1, 2, 255, 256, 65536, 1 << 24]))
        return np.full(n, v, dtype=np.uint32)
    if kind == 1:  # small mixed (bitpack bait for full blocks)
        return np.array(draw(st.lists(st.integers(1, 31), min_size=n,
                                      max_size=n)), dtype=np.uint32)
    if kind == 2:  # byte-width straddle (svb tails)
        return np.array(draw(st.lists(st.sampled_from(
            [1, 255, 256, 65535, 65536, (1 << 24) - 1, 1 << 24]),
            min_size=n, max_size=n)), dtype=np.uint32)
    return np.array(draw(st.lists(st.integers(1, (1 << 31) - 1), min_size=n,
                                  max_size=n)), dtype=np.uint32)


@settings(max_examples=300, deadline=None)
@given(docs=docs_strategy(), prev_gap=st.integers(0, 1 << 20))
def test_doc_block_roundtrip_cross(docs, prev_gap):
    if len(docs) == 0:
        return
    prev = int(docs[0]) - 1 if prev_gap > int(docs[0]) - 1 else prev_gap
    docs = docs + 0  # copy
    if prev >= int(docs[0]):
        prev = int(docs[0]) - 1
    enc_h = sa.encode_doc_block(docs, prev)
    enc_o = po.encode_doc_block(docs, prev)
    assert enc_h == enc_o, "encoder family/bytes diverge (host vs oracle)"
    dec_o, used_o = po.decode_doc_block(enc_h, len(docs), prev)
    np.testing.assert_array_equal(dec_o, docs)
    assert used_o == len(enc_h)
    dec_h, used_h = sa.decode_doc_block(enc_o, len(docs), prev)
    np.testing.assert_array_equal(dec_h, docs)
    assert used_h == len(enc_o)


@settings(max_examples=300, deadline=None)
@given(freqs=freqs_strategy())
def test_freq_block_roundtrip_cross(freqs):
    enc_h = sa.encode_freq_block(freqs)
    enc_o = po.encode_freq_block(freqs)
    assert enc_h == enc_o
    dec_o, used_o = po.decode_freq_block(enc_h, len(freqs))
    np.testing.assert_array_equal(dec_o, freqs)
    assert used_o == len(enc_h)
    dec_h, used_h = sa.decode_freq_block(enc_o, len(freqs))
    np.testing.assert_array_equal(dec_h, freqs)
    assert used_h == len(enc_o)


@settings(max_examples=120, deadline=None)
@given(vals=st.lists(st.integers(-(1 << 31), (1 << 31) - 1), min_size=1,
                     max_size=4000),
       base=st.integers(-(1 << 60), 1 << 60),
       tile=st.sampled_from([1, 7, 1000]))
def test_col_i64_roundtrip_fuzz(vals, base, tile):
    # per-group delta must fit u32 (the round-1 codec contract); the base
    # can sit anywhere in i64
    arr = np.array(vals * tile, dtype=np.int64)[:200_000] + base
    blob = sa.encode_col_i64(arr)
    out = sa.decode_col_i64(blob, len(arr))
    np.testing.assert_array_equal(out, arr)


def test_col_i64_wide_range_rejected():
    # > 32-bit in-group delta range is rejected loudly, not mis-encoded
    arr = np.array([0, 1 << 40], dtype=np.int64)
    with pytest.raises(AssertionError):
        sa.encode_col_i64(arr)


@settings(max_examples=80, deadline=None)
@given(data=st.data())
def test_segment_pipeline_fuzz(data):
    """Whole-segment pipeline on random corpora: build -> oracle decode
    round trip -> exact top-k vs an independent brute-force scorer. Random
    postings shapes cross codec-family boundaries inside one segment."""
    from tests.test_topk import brute_topk  # reuse the independent scorer

    doc_count = data.draw(st.integers(130, 3000))
    nterms = data.draw(st.integers(1, 4))
    rng = np.random.default_rng(data.draw(st.integers(0, 1 << 30)))
    postings = []
    for _ in range(nterms):
        style = data.draw(st.integers(0, 3))
        if style == 0:  # dense run (bitset bait)
            n = min(doc_count, data.draw(st.integers(1, 400)))
            start = rng.integers(1, doc_count - n + 2)
            docs = np.arange(start, start + n, dtype=np.uint32)
        elif style == 1:  # uniform random
            n = data.draw(st.integers(1, min(600, doc_count)))
            docs = np.sort(rng.choice(
                np.arange(1, doc_count + 1, dtype=np.uint32), n,
                replace=False))
        elif style == 2:  # strided (all-same deltas)
            step = data.draw(st.integers(1, 50))
            docs = np.arange(1, doc_count + 1, step, dtype=np.uint32)
        else:  # empty term
            docs = np.zeros(0, dtype=np.uint32)
        freqs = rng.integers(1, 200, len(docs)).astype(np.uint32)
        postings.append((docs, freqs))
    norms = sa.synth_norms(int(rng.integers(0, 1 << 20)), doc_count)
    blob = sa.build_segment(doc_count, postings, norms)
    for t, (docs, freqs) in enumerate(postings):
        if len(docs) == 0:
            continue
        ddocs, dfreqs = po.decode_term(blob, t, len(docs))
        np.testing.assert_array_equal(ddocs, docs)
        np.testing.assert_array_equal(dfreqs, freqs)
    k = data.draw(st.sampled_from([1, 10, 500]))
    mm = data.draw(st.integers(1, nterms))
    hits, total = po.execute_topk([blob], list(range(nterms)),
                                  [1.0] * nterms, k, min_match=mm)
    sels = [0.0] * nterms  # brute_topk ignores sels content beyond length
    order, scores, nmatch = brute_topk(postings, norms, doc_count, sels, k,
                                       min_match=mm)
    assert total == nmatch
    assert [int(h["doc"]) for h in hits] == [int(d) for d in order]
    for h in hits:
        assert h["score"] == scores[int(h["doc"])]


@settings(max_examples=60, deadline=None)
@given(data=st.data())
def test_scan_agg_oracle_fuzz(data):
    """Oracle scan->filter->group-by vs direct numpy on random shapes
    (group counts, predicate ops, value ranges incl. negatives)."""
    rows = data.draw(st.integers(1, 30_000))
    ngroups = data.draw(st.sampled_from([1, 3, 64, 1024]))
    rng = np.random.default_rng(data.draw(st.integers(0, 1 << 30)))
    keys = rng.integers(0, ngroups, rows).astype(np.int64)
    v1 = rng.integers(-(1 << 40), 1 << 40, rows).astype(np.int64)
    v2 = rng.normal(0, 1, rows).astype(np.float32)
    op = data.draw(st.sampled_from([1, 2, 3]))
    lo = int(rng.integers(-(1 << 40), 1 << 40))
    hi = int(rng.integers(lo, 1 << 40)) if op == 3 else 0
    ocnt, osi, osf, opassed = po.scan_agg(keys, v1, v2, ngroups, pred_op=op,
                                          lo=lo, hi=hi)
    if op == 1:
        mask = v1 < lo
    elif op == 2:
        mask = v1 >= lo
    else:
        mask = (v1 >= lo) & (v1 <= hi)
    assert opassed == int(mask.sum())
    np.testing.assert_array_equal(
        ocnt, np.bincount(keys[mask], minlength=ngroups))
    # int64 sums wrap exactly like the oracle's (use object-free wraparound)
    esi = np.zeros(ngroups, dtype=np.int64)
    np.add.at(esi, keys[mask], v1[mask])
    np.testing.assert_array_equal(osi, esi)
    esf = np.bincount(keys[mask], weights=v2[mask].astype(np.float64),
                      minlength=ngroups)
    np.testing.assert_allclose(osf, esf, rtol=1e-7, atol=1e-9)


@settings(max_examples=40, deadline=None)
@given(data=st.data())
def test_hybrid_chain_oracle_fuzz(data):
    """Oracle hybrid predicate chains vs direct numpy: random chain
    shapes (1-3 predicates, mixed LT/GE/BETWEEN) over random columns."""
    doc_count = data.draw(st.integers(2000, 20_000))
    seed = data.draw(st.integers(0, 1 << 30))
    rng = np.random.default_rng(seed)
    sels = [0.15, 0.08]
    postings = [sa.synth_postings(seed % 1000, doc_count, t, s)
                for t, s in enumerate(sels)]
    norms = sa.synth_norms(seed % 1000, doc_count)
    blob = sa.build_segment(doc_count, postings, norms)
    span = 1 << 20
    ncols = data.draw(st.integers(1, 3))
    cols = [rng.integers(0, span, doc_count + 1).astype(np.int64)
            for _ in range(ncols)]
    ops = [3] + [data.draw(st.sampled_from([1, 2, 3]))
                 for _ in range(ncols - 1)]
    los = [int(span * 0.2)] + [int(rng.integers(0, span))
                               for _ in range(ncols - 1)]
    his = []
    for i, op in enumerate(ops):
        his.append(int(rng.integers(los[i], span)) if op == 3 else 0)
    nb = data.draw(st.sampled_from([1, 16, 128]))
    hits, total, bcnt, bsum = po.execute_topk_hybrid_chain(
        blob, [0, 1], [1.0, 1.0], 100, cols, ops, los, his, nb)
    # numpy expectation
    matched = np.zeros(doc_count + 1, dtype=bool)
    for docs, _ in postings:
        matched[docs] = True
    mask = matched.copy()
    for i, op in enumerate(ops):
        c = cols[i]
        if op == 1:
            mask &= c < los[i]
        elif op == 2:
            mask &= c >= los[i]
        else:
            mask &= (c >= los[i]) & (c <= his[i])
    assert total == int(mask.sum())
    surv = np.nonzero(mask)[0]
    c0 = cols[0]
    spanb = his[0] - los[0] + 1
    ecnt = np.zeros(nb, dtype=np.int64)
    esum = np.zeros(nb, dtype=np.int64)
    for d in surv:
        b = min((int(c0[d]) - los[0]) * nb // spanb, nb - 1)
        ecnt[b] += 1
        esum[b] += int(c0[d])
    np.testing.assert_array_equal(bcnt, ecnt)
    np.testing.assert_array_equal(bsum, esum)
    assert set(int(h["doc"]) for h in hits) <= set(int(d) for d in surv)


def test_col_i64_malformed_rejected():
    """table-free validation of the FoR blob: truncation and bad magic are
    caught host-side before any decode."""
    arr = np.arange(1000, dtype=np.int64)
    blob = bytearray(sa.encode_col_i64(arr))
    bad = bytes(blob[:40])  # truncated below header+desc
    with pytest.raises(AssertionError):
        sa.decode_col_i64(bad, 1000)
    blob2 = bytearray(blob)
    blob2[0] ^= 0xFF  # corrupt magic
    with pytest.raises(AssertionError):
        sa.decode_col_i64(bytes(blob2), 1000)
