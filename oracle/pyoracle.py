"""pyoracle — ctypes bindings for the CPU oracle (oracle/liboracle.so) and
the reference's own simdcomp (oracle/_ref/libsimdcomp_ref.so).

TEST INFRASTRUCTURE ONLY: importable from tests/, __graft_entry__.smoke()
and bench.py's cpu_baseline leg, and nowhere else (see sdb_oracle.c header).
"""

import ctypes as C
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
PU32 = C.POINTER(C.c_uint32)
PU64 = C.POINTER(C.c_uint64)


class OScoreDoc(C.Structure):
    _fields_ = [
        ("score", C.c_float),
        ("doc", C.c_uint32),
        ("seg", C.c_uint32),
    ]


class OSegBlob(C.Structure):
    _fields_ = [("blob", C.c_void_p), ("size", C.c_uint64)]


_lib = None
_ref = None


def lib():
    global _lib
    if _lib is None:
        path = os.path.join(_DIR, "liboracle.so")
        if not os.path.exists(path):
            subprocess.run(["make", "-C", _DIR, "liboracle.so"], check=True)
        _lib = C.CDLL(path)
    return _lib


def ref_simdcomp():
    """The reference's own simdcomp, or None when not built (GPU box without
    /root/reference and without a prebuilt _ref)."""
    global _ref
    if _ref is None:
        path = os.path.join(_DIR, "_ref", "libsimdcomp_ref.so")
        if not os.path.exists(path):
            subprocess.run(["make", "-C", _DIR, "ref"], check=False)
        if os.path.exists(path):
            _ref = C.CDLL(path)
    return _ref


def _u32arr(a):
    return np.ascontiguousarray(a, dtype=np.uint32)


def decode_doc_block(payload, length, prev):
    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    consumed = lib().o_decode_doc_block(
        buf.ctypes.data_as(C.POINTER(C.c_uint8)), C.c_uint32(length),
        C.c_uint32(prev), out.ctypes.data_as(PU32))
    return out[:length].copy(), consumed


def decode_freq_block(payload, length):
    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    consumed = lib().o_decode_freq_block(
        buf.ctypes.data_as(C.POINTER(C.c_uint8)), C.c_uint32(length),
        out.ctypes.data_as(PU32))
    return out[:length].copy(), consumed


def encode_doc_block(docs, prev):
    docs = _u32arr(docs)
    out = np.zeros(len(docs) * 5 + 16, dtype=np.uint8)
    size = lib().o_encode_doc_block(
        docs.ctypes.data_as(PU32), C.c_uint32(len(docs)), C.c_uint32(prev),
        out.ctypes.data_as(C.POINTER(C.c_uint8)))
    return bytes(out[:size])


def encode_freq_block(freqs):
    freqs = _u32arr(freqs)
    out = np.zeros(len(freqs) * 5 + 16, dtype=np.uint8)
    size = lib().o_encode_freq_block(
        freqs.ctypes.data_as(PU32), C.c_uint32(len(freqs)),
        out.ctypes.data_as(C.POINTER(C.c_uint8)))
    return bytes(out[:size])


def decode_term(blob, term_idx, df):
    """Decode a whole term's postings from a segment blob via the oracle."""
    buf = np.frombuffer(blob, dtype=np.uint8)

    class _View(C.Structure):
        _fields_ = [("hdr", C.c_void_p), ("terms", C.c_void_p),
                    ("desc", C.c_void_p), ("norms", C.c_void_p),
                    ("payload", C.c_void_p)]

    v = _View()
    rc = lib().o_segment_parse(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(blob)), C.byref(v))
    assert rc == 0, rc
    docs = np.zeros(df, dtype=np.uint32)
    freqs = np.zeros(df, dtype=np.uint32)
    rc = lib().o_decode_term(C.byref(v), C.c_uint32(term_idx),
                             docs.ctypes.data_as(PU32),
                             freqs.ctypes.data_as(PU32))
    assert rc == 0, rc
    return docs, freqs


def bm25_stats(dwf, dwt, ttf, k=1.2, b=0.75):
    idf = C.c_float(0)
    nc = C.c_float(0)
    nl = C.c_float(0)
    lib().o_bm25_stats(C.c_uint64(dwf), C.c_uint64(dwt), C.c_uint64(ttf),
                       C.c_float(k), C.c_float(b), C.byref(idf),
                       C.byref(nc), C.byref(nl))
    return idf.value, nc.value, nl.value


def _mkblobs(blobs):
    bufs = [np.frombuffer(b, dtype=np.uint8) for b in blobs]
    arr = (OSegBlob * len(blobs))()
    for i, b in enumerate(bufs):
        arr[i].blob = b.ctypes.data_as(C.c_void_p).value
        arr[i].size = len(b)
    return arr, bufs  # keep bufs alive


def _hits_to_np(hits, n):
    res = np.zeros(n, dtype=[("score", "f4"), ("doc", "u4"),
                             ("segment", "u4")])
    for i in range(n):
        res[i] = (hits[i].score, hits[i].doc, hits[i].seg)
    return res


SCORERS = {"bm25": 0, "tfidf": 1, "tfidf_norm": 2}


def execute_topk(blobs, term_idx, boosts, k, min_match=1, k1=1.2, b=0.75,
                 global_stats=None, scorer="bm25", filter_boost=None,
                 live_mask=None):
    """EXACT top-k (parity oracle). blobs: list of segment blob bytes.
    filter_boost: optional per-doc f32 multiplier (doc_count+1, 1-based) —
    the HasFilterBoost scorer variants; single-segment only. live_mask:
    optional uint64 bitmap (bit d of word d>>6 = doc d live; seg.mask
    deleted-docs analogue); single-segment only."""
    lib().o_set_scorer(C.c_uint32(SCORERS[scorer]))
    fbkeep = None
    lmkeep = None
    if filter_boost is not None:
        assert len(blobs) == 1, "filter_boost: single segment only"
        fbkeep = np.ascontiguousarray(filter_boost, dtype=np.float32)
        lib().o_set_filter_boost(
            fbkeep.ctypes.data_as(C.POINTER(C.c_float)))
    if live_mask is not None:
        assert len(blobs) == 1, "live_mask: single segment only"
        lmkeep = np.ascontiguousarray(live_mask, dtype=np.uint64)
        lib().o_set_live_mask(
            lmkeep.ctypes.data_as(C.POINTER(C.c_uint64)))
    arr, keep = _mkblobs(blobs)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    g_dwf, g_ttf, g_dwt_ptr, keep2 = 0, 0, None, None
    if global_stats is not None:
        g_dwf, g_ttf, dwt = global_stats
        keep2 = np.ascontiguousarray(dwt, dtype=np.uint64)
        g_dwt_ptr = keep2.ctypes.data_as(PU64)
    hits = (OScoreDoc * k)()
    out_count = C.c_uint32(0)
    total = C.c_uint64(0)
    rc = lib().o_execute_topk(
        arr, C.c_uint32(len(blobs)), ti.ctypes.data_as(PU32),
        bo.ctypes.data_as(C.POINTER(C.c_float)), C.c_uint32(len(ti)),
        C.c_uint32(min_match), C.c_float(k1), C.c_float(b),
        C.c_uint64(g_dwf), g_dwt_ptr, C.c_uint64(g_ttf), C.c_uint32(k),
        hits, C.byref(out_count), C.byref(total))
    if filter_boost is not None:
        lib().o_set_filter_boost(None)
    if live_mask is not None:
        lib().o_set_live_mask(None)
    assert rc == 0, rc
    del keep, keep2, fbkeep, lmkeep
    return _hits_to_np(hits, out_count.value), total.value


def execute_topk_mech(blobs, term_idx, boosts, k, min_match=1, k1=1.2,
                      b=0.75, global_stats=None):
    """2k-buffer/nth_element mechanics emulation (reference fixture
    validation; single-thread timed CPU path)."""
    lib().o_set_scorer(0)  # these entry points are BM25-only
    arr, keep = _mkblobs(blobs)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    g_dwf, g_ttf, g_dwt_ptr, keep2 = 0, 0, None, None
    if global_stats is not None:
        g_dwf, g_ttf, dwt = global_stats
        keep2 = np.ascontiguousarray(dwt, dtype=np.uint64)
        g_dwt_ptr = keep2.ctypes.data_as(PU64)
    hits = (OScoreDoc * (2 * k))()
    out_count = C.c_uint32(0)
    total = C.c_uint64(0)
    rc = lib().o_execute_topk_mech(
        arr, C.c_uint32(len(blobs)), ti.ctypes.data_as(PU32),
        bo.ctypes.data_as(C.POINTER(C.c_float)), C.c_uint32(len(ti)),
        C.c_uint32(min_match), C.c_float(k1), C.c_float(b),
        C.c_uint64(g_dwf), g_dwt_ptr, C.c_uint64(g_ttf), C.c_uint32(k),
        hits, C.byref(out_count), C.byref(total))
    assert rc == 0, rc
    del keep, keep2
    return _hits_to_np(hits, out_count.value), total.value


def execute_topk_mt(blob, term_idx, boosts, k, min_match=1, k1=1.2, b=0.75,
                    nthreads=0, global_stats=None, iters=1):
    """Multithreaded mechanics baseline over one segment (timed CPU leg).
    iters > 1 repeats the whole query inside the thread pool (thread
    creation would otherwise dominate sub-ms queries)."""
    lib().o_set_scorer(0)  # these entry points are BM25-only
    if nthreads == 0:
        nthreads = os.cpu_count() or 1
    buf = np.frombuffer(blob, dtype=np.uint8)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    g_dwf, g_ttf, g_dwt_ptr, keep2 = 0, 0, None, None
    if global_stats is not None:
        g_dwf, g_ttf, dwt = global_stats
        keep2 = np.ascontiguousarray(dwt, dtype=np.uint64)
        g_dwt_ptr = keep2.ctypes.data_as(PU64)
    hits = (OScoreDoc * k)()
    out_count = C.c_uint32(0)
    total = C.c_uint64(0)
    rc = lib().o_execute_topk_mt_iters(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(blob)),
        ti.ctypes.data_as(PU32), bo.ctypes.data_as(C.POINTER(C.c_float)),
        C.c_uint32(len(ti)), C.c_uint32(min_match), C.c_float(k1),
        C.c_float(b), C.c_uint64(g_dwf), g_dwt_ptr, C.c_uint64(g_ttf),
        C.c_uint32(k), C.c_uint32(nthreads), C.c_uint32(iters), hits,
        C.byref(out_count), C.byref(total))
    assert rc == 0, rc
    del keep2
    return _hits_to_np(hits, out_count.value), total.value


def scan_agg(keys, v1, v2, ngroups, pred_op=0, lo=0, hi=0):
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    v1 = np.ascontiguousarray(v1, dtype=np.int64)
    v2 = np.ascontiguousarray(v2, dtype=np.float32)
    cnt = np.zeros(ngroups, dtype=np.int64)
    si = np.zeros(ngroups, dtype=np.int64)
    sf = np.zeros(ngroups, dtype=np.float64)
    passed = C.c_uint64(0)
    PI64 = C.POINTER(C.c_int64)
    rc = lib().o_scan_agg(
        keys.ctypes.data_as(PI64), v1.ctypes.data_as(PI64),
        v2.ctypes.data_as(C.POINTER(C.c_float)), C.c_uint64(len(keys)),
        C.c_uint32(ngroups), C.c_int(pred_op), C.c_int64(lo), C.c_int64(hi),
        cnt.ctypes.data_as(PI64), si.ctypes.data_as(PI64),
        sf.ctypes.data_as(C.POINTER(C.c_double)), C.byref(passed))
    assert rc == 0, rc
    return cnt, si, sf, passed.value


def str_pred_mask(values, op, lo, hi=None):
    """ORACLE (test infrastructure): raw variable-width string predicate,
    memcmp order on unsigned bytes — restates the reference's pushed
    typed compares (index/table_filter_iterator.hpp:104-227) applied to
    var-width string vectors (the reference routes raw strings through
    its DuckDB fork, column_reader.hpp:247-255; parity at result level).
    Python bytes comparison IS lexicographic unsigned-byte order, an
    implementation independent of the GPU kernel's per-lane loop."""
    lob = lo.encode() if isinstance(lo, str) else bytes(lo or b"")
    hib = hi.encode() if isinstance(hi, str) else bytes(hi or b"")
    out = np.zeros(len(values), dtype=bool)
    for i, v in enumerate(values):
        b = v.encode() if isinstance(v, str) else bytes(v)
        if op == "lt":
            m = b < lob
        elif op == "ge":
            m = b >= lob
        elif op == "eq":
            m = b == lob
        elif op == "between":
            m = lob <= b <= hib
        elif op == "prefix":
            m = b.startswith(lob)
        else:
            raise ValueError(op)
        out[i] = m
    return out


def fnv1a64(b):
    """ORACLE (test infrastructure): FNV-1a 64 over bytes — restates the
    device strhash_kernel (sdb_scan.hip) for string-key GROUP BY
    parity."""
    h = 14695981039346656037
    for x in bytes(b):
        h ^= x
        h = (h * 1099511628211) & 0xFFFFFFFFFFFFFFFF
    return h


def fsst_decode(enc, symbols):
    """ORACLE (test infrastructure): FSST-style decode — code c < nsym
    expands to symbols[c]; 255 escapes the next literal byte. Raises on
    malformed streams (out-of-range code, escape at end)."""
    out = bytearray()
    i = 0
    enc = bytes(enc)
    while i < len(enc):
        c = enc[i]
        if c == 255:
            if i + 1 >= len(enc):
                raise ValueError("escape at end of stream")
            out.append(enc[i + 1])
            i += 2
        else:
            if c >= len(symbols):
                raise ValueError(f"code {c} out of range")
            out += symbols[c]
            i += 1
    return bytes(out)


# --- reference simdcomp pins ---------------------------------------------

def ref_pack_d1(prev, values128, bits):
    """simdpackwithoutmaskd1 from the reference's vendored simdcomp."""
    r = ref_simdcomp()
    if r is None:
        return None
    vals = _u32arr(values128).copy()
    assert len(vals) == 128
    out = np.zeros(16 * bits + 16, dtype=np.uint8)
    r.simdpackwithoutmaskd1(C.c_uint32(prev), vals.ctypes.data_as(PU32),
                            out.ctypes.data_as(C.c_void_p),
                            C.c_uint32(bits))
    return bytes(out[: 16 * bits])


def ref_unpack_d1(prev, payload, bits):
    r = ref_simdcomp()
    if r is None:
        return None
    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    r.simdunpackd1(C.c_uint32(prev), buf.ctypes.data_as(C.c_void_p),
                   out.ctypes.data_as(PU32), C.c_uint32(bits))
    return out


def ref_pack(values128, bits):
    """simdpackwithoutmask (non-delta; freq blocks)."""
    r = ref_simdcomp()
    if r is None:
        return None
    vals = _u32arr(values128).copy()
    out = np.zeros(16 * bits + 16, dtype=np.uint8)
    r.simdpackwithoutmask(vals.ctypes.data_as(PU32),
                          out.ctypes.data_as(C.c_void_p), C.c_uint32(bits))
    return bytes(out[: 16 * bits])


def ref_unpack(payload, bits):
    r = ref_simdcomp()
    if r is None:
        return None
    buf = np.frombuffer(bytes(payload), dtype=np.uint8)
    out = np.zeros(128, dtype=np.uint32)
    r.simdunpack(buf.ctypes.data_as(C.c_void_p), out.ctypes.data_as(PU32),
                 C.c_uint32(bits))
    return out


def execute_topk_hybrid(blob, term_idx, boosts, k, col, flo, fhi, nbuckets,
                        min_match=1, k1=1.2, b=0.75, global_stats=None):
    """Hybrid exact top-k: BM25 match AND col BETWEEN [flo,fhi], plus
    per-bucket COUNT/SUM over surviving matches."""
    lib().o_set_scorer(0)  # these entry points are BM25-only
    buf = np.frombuffer(blob, dtype=np.uint8)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    col = np.ascontiguousarray(col, dtype=np.int64)
    g_dwf, g_ttf, g_dwt_ptr, keep2 = 0, 0, None, None
    if global_stats is not None:
        g_dwf, g_ttf, dwt = global_stats
        keep2 = np.ascontiguousarray(dwt, dtype=np.uint64)
        g_dwt_ptr = keep2.ctypes.data_as(PU64)
    bcnt = np.zeros(nbuckets, dtype=np.int64)
    bsum = np.zeros(nbuckets, dtype=np.int64)
    hits = (OScoreDoc * k)()
    out_count = C.c_uint32(0)
    total = C.c_uint64(0)
    PI64 = C.POINTER(C.c_int64)
    rc = lib().o_execute_topk_hybrid(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(buf)),
        ti.ctypes.data_as(PU32), bo.ctypes.data_as(C.POINTER(C.c_float)),
        C.c_uint32(len(ti)), C.c_uint32(min_match), C.c_float(k1),
        C.c_float(b), C.c_uint64(g_dwf), g_dwt_ptr, C.c_uint64(g_ttf),
        C.c_uint32(k), col.ctypes.data_as(PI64), C.c_int64(flo),
        C.c_int64(fhi), C.c_uint32(nbuckets),
        bcnt.ctypes.data_as(PI64), bsum.ctypes.data_as(PI64),
        hits, C.byref(out_count), C.byref(total))
    assert rc == 0, rc
    del keep2
    return _hits_to_np(hits, out_count.value), total.value, bcnt, bsum


def execute_topk_hybrid_chain(blob, term_idx, boosts, k, cols, ops, los,
                              his, nbuckets, min_match=1, k1=1.2, b=0.75):
    """Predicate-chain hybrid: cols[0] carries the BETWEEN bucket span
    (ops[0] must be 3); cols[1..] AND-narrow (1=LT 2=GE 3=BETWEEN)."""
    lib().o_set_scorer(0)  # these entry points are BM25-only
    buf = np.frombuffer(blob, dtype=np.uint8)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    ncols = len(cols)
    keep = [np.ascontiguousarray(c, dtype=np.int64) for c in cols]
    PI64 = C.POINTER(C.c_int64)
    col_ptrs = (PI64 * ncols)(*[c.ctypes.data_as(PI64) for c in keep])
    ops_arr = (C.c_int * ncols)(*[int(o) for o in ops])
    los_arr = np.ascontiguousarray(los, dtype=np.int64)
    his_arr = np.ascontiguousarray(his, dtype=np.int64)
    bcnt = np.zeros(nbuckets, dtype=np.int64)
    bsum = np.zeros(nbuckets, dtype=np.int64)
    hits = (OScoreDoc * k)()
    out_count = C.c_uint32(0)
    total = C.c_uint64(0)
    rc = lib().o_execute_topk_hybrid_chain(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(buf)),
        ti.ctypes.data_as(PU32), bo.ctypes.data_as(C.POINTER(C.c_float)),
        C.c_uint32(len(ti)), C.c_uint32(min_match), C.c_float(k1),
        C.c_float(b), C.c_uint32(k), col_ptrs, ops_arr,
        los_arr.ctypes.data_as(PI64), his_arr.ctypes.data_as(PI64),
        C.c_uint32(ncols), C.c_uint32(nbuckets),
        bcnt.ctypes.data_as(PI64), bsum.ctypes.data_as(PI64),
        hits, C.byref(out_count), C.byref(total))
    assert rc == 0, rc
    return _hits_to_np(hits, out_count.value), total.value, bcnt, bsum


def execute_match_docs(blob, term_idx, boosts, cap, col=None, min_match=1,
                       k1=1.2, b=0.75):
    """Streaming match emission (doc-ascending) + optional column gather."""
    lib().o_set_scorer(0)  # these entry points are BM25-only
    buf = np.frombuffer(blob, dtype=np.uint8)
    ti = _u32arr(term_idx)
    bo = np.ascontiguousarray(boosts, dtype=np.float32)
    docs = np.zeros(cap, dtype=np.uint32)
    PI64 = C.POINTER(C.c_int64)
    col_ptr, col_out, col_out_ptr = None, None, None
    if col is not None:
        col = np.ascontiguousarray(col, dtype=np.int64)
        col_ptr = col.ctypes.data_as(PI64)
        col_out = np.zeros(cap, dtype=np.int64)
        col_out_ptr = col_out.ctypes.data_as(PI64)
    out_n = C.c_uint64(0)
    total = C.c_uint64(0)
    rc = lib().o_execute_match_docs(
        buf.ctypes.data_as(C.c_void_p), C.c_uint64(len(buf)),
        ti.ctypes.data_as(PU32), bo.ctypes.data_as(C.POINTER(C.c_float)),
        C.c_uint32(len(ti)), C.c_uint32(min_match), C.c_float(k1),
        C.c_float(b), col_ptr, col_out_ptr,
        docs.ctypes.data_as(PU32), C.c_uint64(cap), C.byref(out_n),
        C.byref(total))
    assert rc == 0, rc
    n = out_n.value
    return docs[:n], (col_out[:n] if col is not None else None), total.value


def scan_agg_mt(keys, v1, v2, ngroups, pred_op=0, lo=0, hi=0, nthreads=0,
                iters=1):
    """Multithreaded scan baseline (timed CPU leg; deterministic merge)."""
    if nthreads == 0:
        nthreads = os.cpu_count() or 1
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    v1 = np.ascontiguousarray(v1, dtype=np.int64)
    v2 = np.ascontiguousarray(v2, dtype=np.float32)
    cnt = np.zeros(ngroups, dtype=np.int64)
    si = np.zeros(ngroups, dtype=np.int64)
    sf = np.zeros(ngroups, dtype=np.float64)
    passed = C.c_uint64(0)
    PI64 = C.POINTER(C.c_int64)
    rc = lib().o_scan_agg_mt(
        keys.ctypes.data_as(PI64), v1.ctypes.data_as(PI64),
        v2.ctypes.data_as(C.POINTER(C.c_float)), C.c_uint64(len(keys)),
        C.c_uint32(ngroups), C.c_int(pred_op), C.c_int64(lo), C.c_int64(hi),
        C.c_uint32(nthreads), C.c_uint32(iters),
        cnt.ctypes.data_as(PI64), si.ctypes.data_as(PI64),
        sf.ctypes.data_as(C.POINTER(C.c_double)), C.byref(passed))
    assert rc == 0, rc
    return cnt, si, sf, passed.value
