"""serenedb_amd — MI355X-native implementation of SereneDB's search-analytics
hot path (BM25 postings disjunction/conjunction + top-k, fused columnar
scan->filter->hash-aggregate), per /root/repo/BASELINE.json north_star.

Python here is plumbing only (ctypes over the C ABI in include/sdb_gpu.h);
the product is the C++/HIP library. The GPU query path REQUIRES a GPU and
fails loudly otherwise — there is no CPU fallback in this package.
"""

import ctypes as C
import os

_PKG_DIR = os.path.dirname(os.path.abspath(__file__))


class SdbScoreDoc(C.Structure):
    """mirrors irs::ScoreDoc (index/iterators.hpp:93-99)"""
    _fields_ = [
        ("score", C.c_float),
        ("doc", C.c_uint32),
        ("segment_idx", C.c_uint32),
    ]


class SdbTermRef(C.Structure):
    _fields_ = [("term_idx", C.c_uint32), ("boost", C.c_float)]


def _copy_blob(ptr, size):
    """Copy `size` bytes from a C pointer. ctypes.string_at truncates its
    size argument to a C int (a >4 GB segment blob came back mod 2^32);
    ctypes.memmove takes size_t."""
    import numpy as np

    out = np.empty(size, dtype=np.uint8)
    C.memmove(out.ctypes.data, ptr, size)
    return out.tobytes()


def _load(name):
    # SDB_GPU_LIB overrides the GPU library path (same-box A/B of two
    # kernel builds; perf work only, never set in tests/bench defaults)
    if name == "libsdb_gpu.so" and os.environ.get("SDB_GPU_LIB"):
        return C.CDLL(os.environ["SDB_GPU_LIB"])
    path = os.path.join(_PKG_DIR, name)
    if not os.path.exists(path):
        raise ImportError(
            f"{name} not built — run `python -m serenedb_amd.build` "
            f"(or __graft_entry__.build())"
        )
    return C.CDLL(path)


# ---------------------------------------------------------------------------
# host library (index-build side; pure C++; always loadable)
# ---------------------------------------------------------------------------
_host = None


def host():
    global _host
    if _host is None:
        lib = _load("libsdb_host.so")
        lib.sdb_host_encode_doc_block.restype = C.c_int
        lib.sdb_host_decode_doc_block.restype = C.c_int
        lib.sdb_host_encode_freq_block.restype = C.c_int
        lib.sdb_host_decode_freq_block.restype = C.c_int
        lib.sdb_host_build_segment.restype = C.c_int
        lib.sdb_host_build_synth_segment.restype = C.c_int
        lib.sdb_host_synth_postings.restype = C.c_int
        lib.sdb_host_synth_norms.restype = C.c_int
        lib.sdb_host_blob_free.restype = None
        lib.sdb_host_bm25_stats.restype = None
        lib.sdb_host_topk_select.restype = C.c_int
        _host = lib
    return _host


def encode_doc_block(docs, prev):
    import numpy as np

    docs = np.ascontiguousarray(docs, dtype=np.uint32)
    out = np.zeros(len(docs) * 5 + 16, dtype=np.uint8)
    size = C.c_uint32(0)
    rc = host().sdb_host_encode_doc_block(
        docs.ctypes.data_as(C.POINTER(C.c_uint32)), len(docs),
        C.c_uint32(prev), out.ctypes.data_as(C.POINTER(C.c_uint8)),
        C.byref(size))
    assert rc == 0, rc
    return bytes(out[: size.value])


def decode_doc_block(payload, length, prev):
    import numpy as np

    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    consumed = C.c_uint32(0)
    rc = host().sdb_host_decode_doc_block(
        buf.ctypes.data_as(C.POINTER(C.c_uint8)), length, C.c_uint32(prev),
        out.ctypes.data_as(C.POINTER(C.c_uint32)), C.byref(consumed))
    assert rc == 0, rc
    return out[:length].copy(), consumed.value


def encode_freq_block(freqs):
    import numpy as np

    freqs = np.ascontiguousarray(freqs, dtype=np.uint32)
    out = np.zeros(len(freqs) * 5 + 16, dtype=np.uint8)
    size = C.c_uint32(0)
    rc = host().sdb_host_encode_freq_block(
        freqs.ctypes.data_as(C.POINTER(C.c_uint32)), len(freqs),
        out.ctypes.data_as(C.POINTER(C.c_uint8)), C.byref(size))
    assert rc == 0, rc
    return bytes(out[: size.value])


def decode_freq_block(payload, length):
    import numpy as np

    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    consumed = C.c_uint32(0)
    rc = host().sdb_host_decode_freq_block(
        buf.ctypes.data_as(C.POINTER(C.c_uint8)), length,
        out.ctypes.data_as(C.POINTER(C.c_uint32)), C.byref(consumed))
    assert rc == 0, rc
    return out[:length].copy(), consumed.value


def build_segment(doc_count, postings, norms=None):
    """postings: list of (docs_u32_array, freqs_u32_array) per term.
    norms: uint32 array of len doc_count+1 (index 0 unused) or None.
    Returns the serialized segment blob (bytes)."""
    import numpy as np

    nterms = len(postings)
    df = np.array([len(d) for d, _ in postings], dtype=np.uint32)
    doc_arrs = [np.ascontiguousarray(d, dtype=np.uint32) for d, _ in postings]
    frq_arrs = [np.ascontiguousarray(f, dtype=np.uint32) for _, f in postings]
    PU32 = C.POINTER(C.c_uint32)
    doc_ptrs = (PU32 * nterms)(*[a.ctypes.data_as(PU32) for a in doc_arrs])
    frq_ptrs = (PU32 * nterms)(*[a.ctypes.data_as(PU32) for a in frq_arrs])
    norm_ptr = None
    if norms is not None:
        norms = np.ascontiguousarray(norms, dtype=np.uint32)
        assert len(norms) == doc_count + 1
        norm_ptr = norms.ctypes.data_as(PU32)
    blob = C.c_void_p(0)
    size = C.c_uint64(0)
    rc = host().sdb_host_build_segment(
        C.c_uint32(doc_count), C.c_uint32(nterms),
        df.ctypes.data_as(PU32), doc_ptrs, frq_ptrs, norm_ptr,
        C.byref(blob), C.byref(size))
    assert rc == 0, rc
    out = _copy_blob(blob, size.value)
    host().sdb_host_blob_free(blob)
    return out


def build_synth_segment(seed, doc_lo, doc_hi, selectivities):
    import numpy as np

    sel = np.ascontiguousarray(selectivities, dtype=np.float64)
    blob = C.c_void_p(0)
    size = C.c_uint64(0)
    rc = host().sdb_host_build_synth_segment(
        C.c_uint64(seed), C.c_uint32(doc_lo), C.c_uint32(doc_hi),
        C.c_uint32(len(sel)), sel.ctypes.data_as(C.POINTER(C.c_double)),
        C.byref(blob), C.byref(size))
    assert rc == 0, rc
    out = _copy_blob(blob, size.value)
    host().sdb_host_blob_free(blob)
    return out


def synth_postings(seed, doc_count, term, sel):
    import numpy as np

    PU32 = C.POINTER(C.c_uint32)
    df = C.c_uint32(0)
    rc = host().sdb_host_synth_postings(
        C.c_uint64(seed), C.c_uint32(doc_count), C.c_uint32(term),
        C.c_double(sel), None, None, C.byref(df))
    assert rc == 0
    docs = np.zeros(df.value, dtype=np.uint32)
    freqs = np.zeros(df.value, dtype=np.uint32)
    rc = host().sdb_host_synth_postings(
        C.c_uint64(seed), C.c_uint32(doc_count), C.c_uint32(term),
        C.c_double(sel), docs.ctypes.data_as(PU32),
        freqs.ctypes.data_as(PU32), C.byref(df))
    assert rc == 0
    return docs, freqs


def synth_norms(seed, doc_count):
    import numpy as np

    norms = np.zeros(doc_count + 1, dtype=np.uint32)
    rc = host().sdb_host_synth_norms(
        C.c_uint64(seed), C.c_uint32(doc_count),
        norms.ctypes.data_as(C.POINTER(C.c_uint32)))
    assert rc == 0
    return norms


def encode_col_i64(vals, group_rows=65536):
    """FoR/bitpack row-group encoding of an i64 column (+zonemaps)."""
    import numpy as np

    vals = np.ascontiguousarray(vals, dtype=np.int64)
    blob = C.c_void_p(0)
    size = C.c_uint64(0)
    rc = host().sdb_host_encode_col_i64(
        vals.ctypes.data_as(C.POINTER(C.c_int64)), C.c_uint64(len(vals)),
        C.c_uint32(group_rows), C.byref(blob), C.byref(size))
    assert rc == 0, rc
    out = _copy_blob(blob, size.value)
    host().sdb_host_blob_free(blob)
    return out


def decode_col_i64(blob, rows):
    import numpy as np

    buf = np.frombuffer(blob, dtype=np.uint8)
    out = np.zeros(rows, dtype=np.int64)
    rc = host().sdb_host_decode_col_i64(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(buf)),
        out.ctypes.data_as(C.POINTER(C.c_int64)), C.c_uint64(rows))
    assert rc == 0, rc
    return out


def encode_col_str_raw(values):
    """Raw variable-width string column: returns (offsets uint64[n+1],
    blob bytes) for sdb_gpu_table_attach_strcol. Values may be str
    (UTF-8 encoded) or bytes; arbitrary bytes incl. NUL are preserved."""
    import numpy as np

    enc = [v.encode() if isinstance(v, str) else bytes(v) for v in values]
    offsets = np.zeros(len(enc) + 1, dtype=np.uint64)
    np.cumsum([len(b) for b in enc], out=offsets[1:])
    return offsets, b"".join(enc)


def encode_col_str_fsst(values, sample_bytes=1 << 18):
    """FSST-style compression for a raw string column (SURVEY.md 8f row
    3): builds a symbol table (<=254 symbols of 1..8 bytes, greedy by
    gain over a sample), encodes every string as code bytes (255 escapes
    a literal byte), longest-match, deterministic. Returns (offsets
    uint64[n+1] into the ENCODED blob, enc_blob bytes, symbols
    list[bytes]) for GpuContext.attach_strcol_fsst. Decode restatement
    lives in oracle/pyoracle.fsst_decode."""
    import numpy as np
    from collections import Counter

    enc_in = [v.encode() if isinstance(v, str) else bytes(v)
              for v in values]
    # sample substrings, gain = (len-1 or the escape byte saved) * count
    gains = Counter()
    seen = 0
    for b in enc_in:
        if seen > sample_bytes:
            break
        seen += len(b)
        for i in range(len(b)):
            for L in range(1, 9):
                if i + L > len(b):
                    break
                gains[b[i:i + L]] += L - 1 if L > 1 else 0
        for ch in b:
            gains[bytes([ch])] += 1  # 1-byte symbol saves the escape
    ranked = sorted(gains.items(), key=lambda kv: (-kv[1], kv[0]))
    symbols = [sym for sym, g in ranked[:254] if g > 0]
    symbols.sort(key=lambda x: (-len(x), x))  # longest-match by scan
    by_first = {}
    for code, sym in enumerate(symbols):
        by_first.setdefault(sym[0], []).append((sym, code))

    out = bytearray()
    offsets = np.zeros(len(enc_in) + 1, dtype=np.uint64)
    for r, b in enumerate(enc_in):
        i = 0
        while i < len(b):
            best = None
            for sym, code in by_first.get(b[i], ()):
                if b.startswith(sym, i):
                    best = (sym, code)
                    break  # symbols sorted longest-first
            if best:
                out.append(best[1])
                i += len(best[0])
            else:
                out.append(255)
                out.append(b[i])
                i += 1
        offsets[r + 1] = len(out)
    return offsets, bytes(out), symbols


def encode_col_str(values):
    """Dictionary-encode a string column: sorted-unique dictionary + i64
    codes. The SORTED dictionary is what makes string predicates map to
    contiguous code ranges, so the GPU scan path (sdb_gpu_scan_agg over
    i64/FoR columns) covers string GROUP BY / WHERE with zero new device
    code — the dictionary-vector approach of the reference's analytics
    engine (un-vendored DuckDB fork; result-level parity per SURVEY.md
    §8c). Returns (codes int64 array, dictionary list)."""
    import numpy as np

    dictionary = sorted(set(values))
    index = {s: i for i, s in enumerate(dictionary)}
    codes = np.fromiter((index[v] for v in values), dtype=np.int64,
                        count=len(values))
    return codes, dictionary


def str_pred_to_code(dictionary, op, lo=None, hi=None):
    """Translate a string predicate into an (op, ilo, ihi) triple over the
    sorted dictionary's codes, directly usable as an SdbPredSpec:
      op 'eq'      lo               -> BETWEEN [c, c] (empty if absent)
      op 'between' lo, hi inclusive -> BETWEEN code range
      op 'prefix'  lo               -> BETWEEN over the prefix span
      op 'lt' / 'ge' lo             -> LT / GE boundary code
    An unsatisfiable predicate returns BETWEEN (1, 0), which passes no row."""
    import bisect

    n = len(dictionary)

    def left(s):
        return bisect.bisect_left(dictionary, s)

    EMPTY = (3, 1, 0)  # SDB_PRED_BETWEEN with lo > hi: matches nothing
    if op == "eq":
        i = left(lo)
        return (3, i, i) if i < n and dictionary[i] == lo else EMPTY
    if op == "between":
        a = left(lo)
        z = bisect.bisect_right(dictionary, hi) - 1
        return (3, a, z) if a <= z else EMPTY
    if op == "prefix":
        a = left(lo)
        z = bisect.bisect_left(dictionary, lo + "\U0010ffff") - 1
        return (3, a, z) if a <= z else EMPTY
    if op == "lt":
        return (1, left(lo), 0)  # codes < first code >= lo
    if op == "ge":
        return (2, left(lo), 0)
    raise ValueError(op)


def bm25_stats(docs_with_field, docs_with_term, total_term_freq, k=1.2,
               b=0.75):
    idf = C.c_float(0)
    nc = C.c_float(0)
    nl = C.c_float(0)
    host().sdb_host_bm25_stats(
        C.c_uint64(docs_with_field), C.c_uint64(docs_with_term),
        C.c_uint64(total_term_freq), C.c_float(k), C.c_float(b),
        C.byref(idf), C.byref(nc), C.byref(nl))
    return idf.value, nc.value, nl.value


# ---------------------------------------------------------------------------
# GPU library (the product query path; requires a GPU at call time)
# ---------------------------------------------------------------------------
_gpu = None


def gpu():
    global _gpu
    if _gpu is None:
        lib = _load("libsdb_gpu.so")
        lib.sdb_gpu_version.restype = C.c_char_p
        for f in (
            "sdb_gpu_ctx_create", "sdb_gpu_ctx_destroy",
            "sdb_gpu_segment_load", "sdb_gpu_segment_free",
            "sdb_gpu_execute_topk", "sdb_gpu_decode_term",
            "sdb_gpu_table_load", "sdb_gpu_table_free", "sdb_gpu_scan_agg",
            "sdb_gpu_scan_agg_hash",
            "sdb_gpu_segment_attach_column", "sdb_gpu_execute_topk_hybrid",
            "sdb_gpu_segment_attach_livemask",
            "sdb_gpu_execute_topk_batch", "sdb_gpu_table_attach_validity",
            "sdb_gpu_execute_match_docs", "sdb_gpu_execute_count",
        ):
            getattr(lib, f).restype = C.c_int
        _gpu = lib
    return _gpu


class GpuContext:
    """Owns an SdbGpuCtx. Raises RuntimeError (loudly) without a GPU."""

    def __init__(self, device=0):
        self._lib = gpu()
        self._ctx = C.c_void_p(0)
        rc = self._lib.sdb_gpu_ctx_create(device, C.byref(self._ctx))
        if rc != 0:
            raise RuntimeError(
                f"sdb_gpu_ctx_create failed rc={rc} "
                "(no MI355X visible? the GPU path has no CPU fallback)")
        self._segments = []

    def close(self):
        if self._ctx:
            self._lib.sdb_gpu_ctx_destroy(self._ctx)
            self._ctx = C.c_void_p(0)

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def load_segment(self, blob):
        seg = C.c_void_p(0)
        rc = self._lib.sdb_gpu_segment_load(
            self._ctx, blob, C.c_size_t(len(blob)), C.byref(seg))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_segment_load rc={rc}")
        self._segments.append(seg)
        return seg

    SCORERS = {"bm25": 0, "tfidf": 1, "tfidf_norm": 2}

    def _make_plan(self, term_idx, boosts, min_match, k1, b, global_stats,
                   scorer="bm25", wand=False, filter_boost=False):
        class _Plan(C.Structure):
            _fields_ = [
                ("terms", C.POINTER(SdbTermRef)),
                ("nterms", C.c_uint32),
                ("min_match", C.c_uint32),
                ("k1", C.c_float),
                ("b", C.c_float),
                ("scorer", C.c_uint32),
                ("wand", C.c_uint32),
                ("g_docs_with_field", C.c_uint64),
                ("g_total_term_freq", C.c_uint64),
                ("g_docs_with_term", C.POINTER(C.c_uint64)),
                ("filter_boost", C.c_uint32),
            ]

        terms = (SdbTermRef * len(term_idx))(
            *[SdbTermRef(t, float(bo)) for t, bo in zip(term_idx, boosts)])
        plan = _Plan(terms, len(term_idx), min_match, k1, b,
                     self.SCORERS[scorer], 1 if wand else 0, 0, 0, None,
                     1 if filter_boost else 0)
        plan._keep = terms
        if global_stats is not None:
            dwf, ttf, dwt = global_stats
            dwt_arr = (C.c_uint64 * len(dwt))(*[int(x) for x in dwt])
            plan.g_docs_with_field = int(dwf)
            plan.g_total_term_freq = int(ttf)
            plan.g_docs_with_term = dwt_arr
            plan._keep2 = dwt_arr
        return plan

    def execute_topk(self, segs, term_idx, boosts, k, min_match=1, k1=1.2,
                     b=0.75, global_stats=None, scorer="bm25", wand=False,
                     filter_boost=False):
        """global_stats: optional (docs_with_field, total_term_freq,
        [docs_with_term per term]) for sharded execution. wand=True enables
        exact block-max pruning (OR plans only; total_matches then counts
        visited matches only). filter_boost=True multiplies every term
        contribution by the segment's attached boost column
        (attach_boost; the HasFilterBoost scorer variants)."""
        import numpy as np

        plan = self._make_plan(term_idx, boosts, min_match, k1, b,
                               global_stats, scorer, wand, filter_boost)
        seg_arr = (C.c_void_p * len(segs))(*[C.c_void_p(s.value) for s in segs])
        hits = (SdbScoreDoc * k)()
        out_count = C.c_uint32(0)
        total = C.c_uint64(0)
        rc = self._lib.sdb_gpu_execute_topk(
            self._ctx, seg_arr, len(segs), C.byref(plan), C.c_uint32(k),
            hits, C.byref(out_count), C.byref(total))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_topk rc={rc}")
        n = out_count.value
        dt = np.dtype([("score", "f4"), ("doc", "u4"), ("segment", "u4")])
        res = np.frombuffer(C.string_at(hits, C.sizeof(SdbScoreDoc) * n),
                            dtype=dt).copy() if n else np.zeros(0, dtype=dt)
        return res, total.value

    def attach_column(self, seg, col, slot=0):
        import numpy as np

        col = np.ascontiguousarray(col, dtype=np.int64)
        rc = self._lib.sdb_gpu_segment_attach_column_slot(
            self._ctx, seg, C.c_uint32(slot),
            col.ctypes.data_as(C.POINTER(C.c_int64)))
        if rc != 0:
            raise RuntimeError(f"attach_column rc={rc}")

    def attach_livemask(self, seg, mask):
        """Attach (or with mask=None detach) a live-document bitmap —
        the deleted-docs mask the reference wraps around every scan
        (seg.mask(it), duckdb_search_full_scan.cpp:1898). uint64 words,
        bit d of word d>>6 = doc d live."""
        import numpy as np

        if mask is None:
            rc = self._lib.sdb_gpu_segment_attach_livemask(self._ctx, seg,
                                                           None)
        else:
            m = np.ascontiguousarray(mask, dtype=np.uint64)
            rc = self._lib.sdb_gpu_segment_attach_livemask(
                self._ctx, seg, m.ctypes.data_as(C.POINTER(C.c_uint64)))
        if rc != 0:
            raise RuntimeError(f"attach_livemask rc={rc}")

    def attach_boost(self, seg, boost):
        """Attach the per-doc f32 filter-boost column (doc_count+1,
        1-based docs; bm25.cpp HasFilterBoost variants)."""
        import numpy as np

        boost = np.ascontiguousarray(boost, dtype=np.float32)
        rc = self._lib.sdb_gpu_segment_attach_boost(
            self._ctx, seg, boost.ctypes.data_as(C.POINTER(C.c_float)))
        if rc != 0:
            raise RuntimeError(f"attach_boost rc={rc}")

    def execute_topk_hybrid(self, segs, term_idx, boosts, k, flo, fhi,
                            nbuckets, min_match=1, k1=1.2, b=0.75,
                            global_stats=None):
        import numpy as np

        plan = self._make_plan(term_idx, boosts, min_match, k1, b,
                               global_stats)
        seg_arr = (C.c_void_p * len(segs))(
            *[C.c_void_p(s.value) for s in segs])
        hits = (SdbScoreDoc * k)()
        out_count = C.c_uint32(0)
        total = C.c_uint64(0)
        bcnt = np.zeros(nbuckets, dtype=np.int64)
        bsum = np.zeros(nbuckets, dtype=np.int64)
        PI64 = C.POINTER(C.c_int64)
        rc = self._lib.sdb_gpu_execute_topk_hybrid(
            self._ctx, seg_arr, len(segs), C.byref(plan), C.c_uint32(k),
            C.c_int64(flo), C.c_int64(fhi), C.c_uint32(nbuckets),
            bcnt.ctypes.data_as(PI64), bsum.ctypes.data_as(PI64),
            hits, C.byref(out_count), C.byref(total))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_topk_hybrid rc={rc}")
        n = out_count.value
        dt = np.dtype([("score", "f4"), ("doc", "u4"), ("segment", "u4")])
        res = np.frombuffer(C.string_at(hits, C.sizeof(SdbScoreDoc) * n),
                            dtype=dt).copy() if n else np.zeros(0, dtype=dt)
        return res, total.value, bcnt, bsum

    def execute_topk_hybrid_batch(self, segs, term_idx, boosts, k, flo,
                                  fhi, nbuckets, nq, min_match=1, k1=1.2,
                                  b=0.75, global_stats=None,
                                  all_hits=False):
        """Pipelined batch of nq identical hybrid queries (each fully
        re-executed; per-query semantics == execute_topk_hybrid).
        Returns (hits, totals[nq], bcnt [nq, nbuckets], bsum [nq,
        nbuckets])."""
        import numpy as np

        plan = self._make_plan(term_idx, boosts, min_match, k1, b,
                               global_stats)
        seg_arr = (C.c_void_p * len(segs))(
            *[C.c_void_p(s.value) for s in segs])
        hits = (SdbScoreDoc * (nq * k))()
        counts = (C.c_uint32 * nq)()
        totals = (C.c_uint64 * nq)()
        bcnt = np.zeros((nq, nbuckets), dtype=np.int64)
        bsum = np.zeros((nq, nbuckets), dtype=np.int64)
        PI64 = C.POINTER(C.c_int64)
        rc = self._lib.sdb_gpu_execute_topk_hybrid_batch(
            self._ctx, seg_arr, len(segs), C.byref(plan), C.c_uint32(k),
            C.c_int64(flo), C.c_int64(fhi), C.c_uint32(nbuckets),
            C.c_uint32(nq), bcnt.ctypes.data_as(PI64),
            bsum.ctypes.data_as(PI64), hits, counts, totals)
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_topk_hybrid_batch rc={rc}")
        dt = np.dtype([("score", "f4"), ("doc", "u4"), ("segment", "u4")])

        def conv(q):
            n = counts[q]
            base = C.addressof(hits) + q * k * C.sizeof(SdbScoreDoc)
            return np.frombuffer(C.string_at(base, 12 * n), dtype=dt,
                                 count=n).copy()

        out = ([conv(q) for q in range(nq)] if all_hits else conv(nq - 1))
        return out, [int(totals[q]) for q in range(nq)], bcnt, bsum

    def execute_topk_hybrid_chain(self, segs, term_idx, boosts, k, preds,
                                  nbuckets, min_match=1, k1=1.2, b=0.75,
                                  global_stats=None):
        """preds: list of (slot, op, lo, hi); preds[0] must be BETWEEN
        (op=3) and defines the bucket span (ColFilterChain semantics,
        table_filter_iterator.hpp:104-312)."""
        import numpy as np

        class _HP(C.Structure):
            _fields_ = [("slot", C.c_uint32), ("op", C.c_int),
                        ("lo", C.c_int64), ("hi", C.c_int64)]

        plan = self._make_plan(term_idx, boosts, min_match, k1, b,
                               global_stats)
        seg_arr = (C.c_void_p * len(segs))(
            *[C.c_void_p(s.value) for s in segs])
        parr = (_HP * len(preds))(
            *[_HP(s, o, int(lo), int(hi)) for s, o, lo, hi in preds])
        hits = (SdbScoreDoc * k)()
        out_count = C.c_uint32(0)
        total = C.c_uint64(0)
        bcnt = np.zeros(nbuckets, dtype=np.int64)
        bsum = np.zeros(nbuckets, dtype=np.int64)
        PI64 = C.POINTER(C.c_int64)
        rc = self._lib.sdb_gpu_execute_topk_hybrid_chain(
            self._ctx, seg_arr, len(segs), C.byref(plan), C.c_uint32(k),
            parr, C.c_uint32(len(preds)), C.c_uint32(nbuckets),
            bcnt.ctypes.data_as(PI64), bsum.ctypes.data_as(PI64),
            hits, C.byref(out_count), C.byref(total))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_topk_hybrid_chain rc={rc}")
        n = out_count.value
        dt = np.dtype([("score", "f4"), ("doc", "u4"), ("segment", "u4")])
        res = np.frombuffer(C.string_at(hits, C.sizeof(SdbScoreDoc) * n),
                            dtype=dt).copy() if n else np.zeros(0, dtype=dt)
        return res, total.value, bcnt, bsum

    def execute_count(self, segs, term_idx, boosts, min_match=1, k1=1.2,
                      b=0.75):
        plan = self._make_plan(term_idx, boosts, min_match, k1, b, None)
        seg_arr = (C.c_void_p * len(segs))(
            *[C.c_void_p(s.value) for s in segs])
        total = C.c_uint64(0)
        rc = self._lib.sdb_gpu_execute_count(
            self._ctx, seg_arr, len(segs), C.byref(plan), C.byref(total))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_count rc={rc}")
        return total.value

    def execute_match_docs(self, seg, term_idx, boosts, cap, min_match=1,
                           k1=1.2, b=0.75, with_col=False):
        import numpy as np

        plan = self._make_plan(term_idx, boosts, min_match, k1, b, None)
        docs = np.zeros(cap, dtype=np.uint32)
        cols = np.zeros(cap, dtype=np.int64) if with_col else None
        out_n = C.c_uint64(0)
        total = C.c_uint64(0)
        rc = self._lib.sdb_gpu_execute_match_docs(
            self._ctx, seg, C.byref(plan),
            docs.ctypes.data_as(C.POINTER(C.c_uint32)),
            cols.ctypes.data_as(C.POINTER(C.c_int64)) if with_col else None,
            C.c_uint64(cap), C.byref(out_n), C.byref(total))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_match_docs rc={rc}")
        n = out_n.value
        return docs[:n], (cols[:n] if with_col else None), total.value

    def execute_match_docs_multi(self, segs, term_idx, boosts, cap,
                                 min_match=1, k1=1.2, b=0.75,
                                 with_col=False):
        """Streaming scan over several resident segments: the reference's
        RunStreamingScan worker loop claims segments one at a time
        (duckdb_search_full_scan.cpp:2370 via g.next_segment), so the
        multi-segment form IS a per-segment loop. Emits (segment_idx, doc)
        pairs segment-ascending then doc-ascending, column values gathered
        per segment when attached."""
        import numpy as np

        seg_out, doc_out, col_out = [], [], []
        total = 0
        left = cap
        for si, seg in enumerate(segs):
            docs, cols, t = self.execute_match_docs(
                seg, term_idx, boosts, left, min_match=min_match, k1=k1,
                b=b, with_col=with_col)
            total += t
            seg_out.append(np.full(len(docs), si, dtype=np.uint32))
            doc_out.append(docs)
            if with_col:
                col_out.append(cols)
            left = max(left - len(docs), 0)  # keep counting total_matches
                                             # in later segments past cap
        segcat = np.concatenate(seg_out) if seg_out else np.zeros(0, "u4")
        doccat = np.concatenate(doc_out) if doc_out else np.zeros(0, "u4")
        colcat = (np.concatenate(col_out) if with_col and col_out else None)
        return segcat, doccat, colcat, total

    def decode_term(self, seg, term_idx, df):
        import numpy as np

        docs = np.zeros(df, dtype=np.uint32)
        freqs = np.zeros(df, dtype=np.uint32)
        PU32 = C.POINTER(C.c_uint32)
        rc = self._lib.sdb_gpu_decode_term(
            self._ctx, seg, C.c_uint32(term_idx),
            docs.ctypes.data_as(PU32), freqs.ctypes.data_as(PU32))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_decode_term rc={rc}")
        return docs, freqs

    # ---- columnar scan helpers (ctypes mirrors of sdb_gpu.h) ----

    class _ColView(C.Structure):
        _fields_ = [("data", C.c_void_p), ("rows", C.c_uint64),
                    ("type", C.c_int)]

    class _PredSpec(C.Structure):
        _fields_ = [("col", C.c_uint32), ("op", C.c_int),
                    ("ilo", C.c_int64), ("ihi", C.c_int64),
                    ("flo", C.c_float), ("fhi", C.c_float)]

    class _AggSpec(C.Structure):
        _fields_ = [("col", C.c_uint32), ("op", C.c_int)]

    class _AggResult(C.Structure):
        _fields_ = [("i64", C.c_int64), ("f64", C.c_double)]

    def load_table(self, arrays, codecs=None):
        """arrays: list of numpy arrays (int64 or float32). codecs: per
        column, "raw" or "for" (i64 only). Returns an opaque table handle.
        Keeps the encoded blobs alive for the duration of the call only
        (table_load uploads them)."""
        import numpy as np

        n = len(arrays)
        codecs = codecs or ["raw"] * n
        views = (self._ColView * n)()
        keep = []
        rows = len(arrays[0])
        for i, (a, enc) in enumerate(zip(arrays, codecs)):
            if a.dtype == np.float32:
                a = np.ascontiguousarray(a)
                keep.append(a)
                views[i] = self._ColView(
                    a.ctypes.data_as(C.c_void_p).value, rows, 1)
            elif enc == "for":
                blob = encode_col_i64(np.ascontiguousarray(a, np.int64))
                bv = np.frombuffer(blob, dtype=np.uint8)
                keep.append(bv)
                views[i] = self._ColView(
                    bv.ctypes.data_as(C.c_void_p).value, rows, 2)
            else:
                a = np.ascontiguousarray(a, np.int64)
                keep.append(a)
                views[i] = self._ColView(
                    a.ctypes.data_as(C.c_void_p).value, rows, 0)
        tab = C.c_void_p(0)
        rc = self._lib.sdb_gpu_table_load(self._ctx, views, C.c_uint32(n),
                                          C.c_uint64(rows), C.byref(tab))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_table_load rc={rc}")
        return tab

    def free_table(self, tab):
        self._lib.sdb_gpu_table_free(self._ctx, tab)

    def attach_validity(self, tab, col, bits):
        """Attach (mask=None detaches) a column validity bitmap: bit r of
        word r>>6 set = row r valid (NOT NULL). SQL three-valued logic:
        comparisons drop NULL rows, ISNULL/NOTNULL (ops 5/6) evaluate the
        plane alone, SUM skips NULLs, COUNT(*) counts rows."""
        import numpy as np

        if bits is None:
            rc = self._lib.sdb_gpu_table_attach_validity(
                self._ctx, tab, C.c_uint32(col), None)
        else:
            b = np.ascontiguousarray(bits, dtype=np.uint64)
            rc = self._lib.sdb_gpu_table_attach_validity(
                self._ctx, tab, C.c_uint32(col),
                b.ctypes.data_as(C.POINTER(C.c_uint64)))
        if rc != 0:
            raise RuntimeError(f"attach_validity rc={rc}")

    def attach_strcol(self, tab, slot, offsets, blob):
        """Attach a raw variable-width string column (non-dictionary;
        include/sdb_gpu.h sdb_gpu_table_attach_strcol): offsets[rows+1]
        uint64 byte offsets into blob (bytes). Slot 0..3, independent of
        the i64/f32 column list."""
        import numpy as np

        off = np.ascontiguousarray(offsets, dtype=np.uint64)
        bl = np.frombuffer(bytes(blob), dtype=np.uint8) if len(blob) \
            else np.zeros(0, dtype=np.uint8)
        rc = self._lib.sdb_gpu_table_attach_strcol(
            self._ctx, tab, C.c_uint32(slot),
            off.ctypes.data_as(C.POINTER(C.c_uint64)),
            bl.ctypes.data_as(C.POINTER(C.c_uint8)),
            C.c_uint64(len(bl)))
        if rc != 0:
            raise RuntimeError(f"attach_strcol rc={rc}")

    def attach_strcol_fsst(self, tab, slot, offsets, enc_blob, symbols):
        """Attach an FSST-style compressed string column (see
        encode_col_str_fsst). Predicates via the same strpred_mask call
        decode on the fly — semantics identical to the raw slot."""
        import numpy as np

        off = np.ascontiguousarray(offsets, dtype=np.uint64)
        bl = np.frombuffer(bytes(enc_blob), dtype=np.uint8) if \
            len(enc_blob) else np.zeros(0, dtype=np.uint8)
        sym_off = np.zeros(len(symbols) + 1, dtype=np.uint32)
        np.cumsum([len(x) for x in symbols], out=sym_off[1:])
        sym = np.frombuffer(b"".join(bytes(x) for x in symbols),
                            dtype=np.uint8) if symbols else \
            np.zeros(0, dtype=np.uint8)
        rc = self._lib.sdb_gpu_table_attach_strcol_fsst(
            self._ctx, tab, C.c_uint32(slot),
            off.ctypes.data_as(C.POINTER(C.c_uint64)),
            bl.ctypes.data_as(C.POINTER(C.c_uint8)), C.c_uint64(len(bl)),
            sym.ctypes.data_as(C.POINTER(C.c_uint8)),
            sym_off.ctypes.data_as(C.POINTER(C.c_uint32)),
            C.c_uint32(len(symbols)))
        if rc != 0:
            raise RuntimeError(f"attach_strcol_fsst rc={rc}")

    STR_OPS = {"lt": 1, "ge": 2, "between": 3, "eq": 4, "prefix": 8}

    def strpred_mask(self, tab, slot, op, lo, hi=None):
        """Evaluate a string predicate over an attached raw string slot
        into its device row bitmask (memcmp order on UTF-8 bytes). op:
        "lt"/"ge"/"between"/"eq"/"prefix" or the numeric SdbPredOp. The
        mask is then consumed by scan_agg / scan_agg_hash via the pred
        tuple (slot, 7, 0, 0)  [op 7 = SDB_PRED_STRMASK]."""
        opn = self.STR_OPS[op] if isinstance(op, str) else int(op)
        lob = lo.encode() if isinstance(lo, str) else bytes(lo or b"")
        hib = hi.encode() if isinstance(hi, str) else bytes(hi or b"")
        la = (C.c_uint8 * max(len(lob), 1))(*lob)
        ha = (C.c_uint8 * max(len(hib), 1))(*hib)
        rc = self._lib.sdb_gpu_strpred_mask(
            self._ctx, tab, C.c_uint32(slot), C.c_int(opn), la,
            C.c_uint32(len(lob)), ha, C.c_uint32(len(hib)))
        if rc != 0:
            raise RuntimeError(f"strpred_mask rc={rc}")

    def scan_agg(self, tab, group_col, ngroups, preds, aggs):
        """preds: list of (col, op, lo, hi) with SdbPredOp numeric op
        (1=LT 2=GE 3=BETWEEN 4=EQ); lo/hi int or float (float for f32
        predicate columns). aggs: list of (col, op) with 0=COUNT
        1=SUM_I64 2=SUM_F64. Returns (i64 results [ngroups, naggs],
        f64 results [ngroups, naggs], rows_passed)."""
        import numpy as np

        np_ = len(preds)
        pa = (self._PredSpec * max(np_, 1))()
        for i, (col, op, lo, hi) in enumerate(preds):
            if isinstance(lo, float) or isinstance(hi, float):
                pa[i] = self._PredSpec(col, op, 0, 0, lo, hi)
            else:
                pa[i] = self._PredSpec(col, op, lo, hi, 0, 0)
        na = len(aggs)
        aa = (self._AggSpec * na)(*[self._AggSpec(c, o) for c, o in aggs])
        out = (self._AggResult * (ngroups * na))()
        passed = C.c_uint64(0)
        rc = self._lib.sdb_gpu_scan_agg(
            self._ctx, tab, C.c_uint32(group_col), C.c_uint32(ngroups), pa,
            C.c_uint32(np_), aa, C.c_uint32(na), out, C.byref(passed))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_scan_agg rc={rc}")
        i64 = np.array([[out[g * na + q].i64 for q in range(na)]
                        for g in range(ngroups)], dtype=np.int64)
        f64 = np.array([[out[g * na + q].f64 for q in range(na)]
                        for g in range(ngroups)], dtype=np.float64)
        return i64, f64, passed.value

    def scan_agg_hash_str(self, tab, str_slot, max_groups, preds, aggs,
                          values=None):
        """GROUP BY string keys over an attached raw/FSST string slot.
        Returns (keys, i64 [n, naggs], f64 [n, naggs], rows_passed):
        keys are the FNV-1a 64 group hashes (signed-i64 ascending), or —
        when `values` (the column's strings) is given — the resolved
        key strings, with hash injectivity over the distinct strings
        VERIFIED (raises on a collision; exactness is never silent)."""
        import numpy as np
        from oracle.pyoracle import fnv1a64

        np_ = len(preds)
        pa = (self._PredSpec * max(np_, 1))()
        for i, (col, op, lo, hi) in enumerate(preds):
            if isinstance(lo, float) or isinstance(hi, float):
                pa[i] = self._PredSpec(col, op, 0, 0, lo, hi)
            else:
                pa[i] = self._PredSpec(col, op, lo, hi, 0, 0)
        na = len(aggs)
        aa = (self._AggSpec * na)(*[self._AggSpec(c, o) for c, o in aggs])
        keys = np.zeros(max_groups, dtype=np.int64)
        out = (self._AggResult * (max_groups * na))()
        ng = C.c_uint64(0)
        passed = C.c_uint64(0)
        rc = self._lib.sdb_gpu_scan_agg_hash_str(
            self._ctx, tab, C.c_uint32(str_slot), C.c_uint64(max_groups),
            pa, C.c_uint32(np_), aa, C.c_uint32(na),
            keys.ctypes.data_as(C.POINTER(C.c_int64)), out, C.byref(ng),
            C.byref(passed))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_scan_agg_hash_str rc={rc}")
        n = ng.value
        i64 = np.array([[out[g * na + q].i64 for q in range(na)]
                        for g in range(n)], dtype=np.int64)
        f64 = np.array([[out[g * na + q].f64 for q in range(na)]
                        for g in range(n)], dtype=np.float64)
        kh = keys[:n]
        if values is not None:
            by_hash = {}
            for v in values:
                b = v.encode() if isinstance(v, str) else bytes(v)
                h = fnv1a64(b)
                if by_hash.setdefault(h, b) != b:
                    raise RuntimeError(
                        f"FNV-1a collision between {by_hash[h]!r} and "
                        f"{b!r}: string group-by would merge groups")
            kh = [by_hash[int(h) & 0xFFFFFFFFFFFFFFFF] for h in kh]
        return kh, i64, f64, passed.value

    def load_table_i64(self, keys, vals, codec="raw"):
        return self.load_table([keys, vals], [codec, codec])

    def scan_agg_count_sum(self, tab, ngroups):
        """COUNT(*) + SUM(col1) grouped by col0, no predicate."""
        return self.scan_agg(tab, 0, ngroups, [],
                             [(0, 0), (1, 1)])

    def scan_agg_hash(self, tab, group_col, max_groups, preds, aggs):
        """General hash aggregate (arbitrary i64 keys). Returns
        (keys[n], i64 results [n, naggs], f64 results [n, naggs],
        rows_passed), rows sorted by key ascending."""
        import numpy as np

        np_ = len(preds)
        pa = (self._PredSpec * max(np_, 1))()
        for i, (col, op, lo, hi) in enumerate(preds):
            if isinstance(lo, float) or isinstance(hi, float):
                pa[i] = self._PredSpec(col, op, 0, 0, lo, hi)
            else:
                pa[i] = self._PredSpec(col, op, lo, hi, 0, 0)
        na = len(aggs)
        aa = (self._AggSpec * na)(*[self._AggSpec(c, o) for c, o in aggs])
        keys_out = np.zeros(max_groups, dtype=np.int64)
        out = (self._AggResult * (max_groups * na))()
        ng = C.c_uint64(0)
        passed = C.c_uint64(0)
        rc = self._lib.sdb_gpu_scan_agg_hash(
            self._ctx, tab, C.c_uint32(group_col), C.c_uint64(max_groups),
            pa, C.c_uint32(np_), aa, C.c_uint32(na),
            keys_out.ctypes.data_as(C.POINTER(C.c_int64)), out,
            C.byref(ng), C.byref(passed))
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_scan_agg_hash rc={rc}")
        n = ng.value
        i64 = np.array([[out[g * na + q].i64 for q in range(na)]
                        for g in range(n)], dtype=np.int64)
        f64 = np.array([[out[g * na + q].f64 for q in range(na)]
                        for g in range(n)], dtype=np.float64)
        return keys_out[:n], i64.reshape(n, na), f64.reshape(n, na), \
            passed.value


    def execute_topk_batch(self, segs, term_idx, boosts, k, nq,
                           min_match=1, k1=1.2, b=0.75, global_stats=None,
                           scorer="bm25", wand=False, filter_boost=False,
                           all_hits=False):
        """Pipelined batch of nq identical-plan queries (each fully
        re-executed); returns (hits — list of nq structured arrays when
        all_hits else just the last query's array — and totals [nq])."""
        import numpy as np

        plan = self._make_plan(term_idx, boosts, min_match, k1, b,
                               global_stats, scorer, wand, filter_boost)
        seg_arr = (C.c_void_p * len(segs))(*[C.c_void_p(s.value)
                                             for s in segs])
        hits = (SdbScoreDoc * (nq * k))()
        counts = (C.c_uint32 * nq)()
        totals = (C.c_uint64 * nq)()
        rc = self._lib.sdb_gpu_execute_topk_batch(
            self._ctx, seg_arr, C.c_uint32(len(segs)), C.byref(plan),
            C.c_uint32(k), C.c_uint32(nq), hits, counts, totals)
        if rc != 0:
            raise RuntimeError(f"sdb_gpu_execute_topk_batch rc={rc}")

        def conv(q):
            n = counts[q]
            base = C.addressof(hits) + q * k * C.sizeof(SdbScoreDoc)
            res = np.frombuffer(
                C.string_at(base, C.sizeof(SdbScoreDoc) * n),
                dtype=[("score", "f4"), ("doc", "u4"), ("segment", "u4")],
                count=n).copy()
            return res

        out = ([conv(q) for q in range(nq)] if all_hits else conv(nq - 1))
        return out, [int(totals[q]) for q in range(nq)]


def ingest_doc(doc_file, term_metas, doc_count, norms=None, has_freq=True):
    """Read a reference-format `.doc` postings file (+ sidecar term metas:
    list of dicts with docs_count, total_freq, doc_start, e_single_doc,
    e_skip_start) and rebuild it as a segment blob. norms: uint32
    [doc_count+1] or None."""
    import numpy as np

    class _TM(C.Structure):
        _fields_ = [("docs_count", C.c_uint32), ("total_freq", C.c_uint64),
                    ("doc_start", C.c_uint64), ("e_single_doc", C.c_uint32),
                    ("e_skip_start", C.c_uint64)]

    n = len(term_metas)
    tm = (_TM * n)(*[_TM(m["docs_count"], m.get("total_freq", 0),
                         m.get("doc_start", 0), m.get("e_single_doc", 0),
                         m.get("e_skip_start", 0)) for m in term_metas])
    buf = np.frombuffer(bytes(doc_file), dtype=np.uint8)
    norm_ptr = None
    if norms is not None:
        norms = np.ascontiguousarray(norms, dtype=np.uint32)
        norm_ptr = norms.ctypes.data_as(C.POINTER(C.c_uint32))
    blob = C.c_void_p(0)
    size = C.c_uint64(0)
    h = host()
    h.sdb_host_ingest_doc.restype = C.c_int
    rc = h.sdb_host_ingest_doc(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(buf)), tm,
        C.c_uint32(n), C.c_uint32(doc_count),
        C.c_uint32(1 if has_freq else 0), norm_ptr,
        C.byref(blob), C.byref(size))
    if rc != 0:
        raise ValueError(f"sdb_host_ingest_doc rc={rc}")
    out = _copy_blob(blob, size.value)
    h.sdb_host_blob_free(blob)
    return out
