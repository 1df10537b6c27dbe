"""Dictionary-encoded string columns (string/dict keys).

The sorted dictionary maps string predicates onto contiguous code ranges;
the scan path then runs unchanged over i64 codes. CPU side: the code
ranges must reproduce direct string comparison exactly, and the oracle
scan over codes must equal a direct string-space evaluation.
"""
import numpy as np
import pytest

import serenedb_amd as sa
from oracle import pyoracle as po


def make_strings(seed, n):
    rng = np.random.default_rng(seed)
    vocab = [f"{c}{i:03d}" for c in "abcdefgh" for i in range(64)]
    return [vocab[i] for i in rng.integers(0, len(vocab), n)], vocab


def apply_code_pred(codes, pred):
    op, lo, hi = pred
    if op == 1:
        return codes < lo
    if op == 2:
        return codes >= lo
    return (codes >= lo) & (codes <= hi)


def test_str_pred_ranges_match_string_space():
    vals, _ = make_strings(1, 5000)
    codes, d = sa.encode_col_str(vals)
    sv = np.array(vals)
    # round trip
    assert [d[c] for c in codes[:100]] == vals[:100]
    cases = [
        (("eq", "c017", None), sv == "c017"),
        (("eq", "zzzz", None), np.zeros(len(sv), bool)),       # absent
        (("between", "b000", "c999"), (sv >= "b000") & (sv <= "c999")),
        (("between", "x", "y"), np.zeros(len(sv), bool)),      # empty
        (("prefix", "d0", None), np.char.startswith(sv, "d0")),
        (("prefix", "q", None), np.zeros(len(sv), bool)),
        (("lt", "c000", None), sv < "c000"),
        (("ge", "f031", None), sv >= "f031"),
        (("lt", "a000", None), np.zeros(len(sv), bool)),       # below min
        (("ge", "a000", None), np.ones(len(sv), bool)),        # all
    ]
    for (op, lo, hi), expect in cases:
        pred = sa.str_pred_to_code(d, op, lo, hi)
        np.testing.assert_array_equal(apply_code_pred(codes, pred), expect,
                                      err_msg=f"{op} {lo} {hi}")


def test_str_groupby_scan_oracle():
    """GROUP BY string key with a string prefix predicate, evaluated by the
    oracle scan over codes, vs direct string-space numpy aggregation."""
    n = 200_000
    keys_s, _ = make_strings(2, n)
    filt_s, _ = make_strings(3, n)
    v2 = np.random.default_rng(4).normal(0, 1, n).astype(np.float32)
    kcodes, kdict = sa.encode_col_str(keys_s)
    fcodes, fdict = sa.encode_col_str(filt_s)
    op, lo, hi = sa.str_pred_to_code(fdict, "prefix", "c")
    assert op == 3
    ngroups = len(kdict)
    ocnt, osi, osf, opassed = po.scan_agg(kcodes, fcodes, v2, ngroups,
                                          pred_op=3, lo=lo, hi=hi)
    fs = np.array(filt_s)
    mask = np.char.startswith(fs, "c")
    assert opassed == int(mask.sum())
    ks = np.array(keys_s)
    for gi, gname in enumerate(kdict):
        sel = mask & (ks == gname)
        assert ocnt[gi] == int(sel.sum())
    # SUM over the filter column's codes is also exact vs string space
    exp_si = np.array([int(fcodes[mask & (ks == g)].sum()) for g in kdict])
    np.testing.assert_array_equal(osi, exp_si)


@pytest.mark.gpu
def test_str_groupby_scan_gpu():
    """String GROUP BY + prefix predicate on the GPU scan path (codes as
    dense i64 AND as FoR/bitpack), vs direct string-space numpy."""
    import ctypes as CT

    ctx = sa.GpuContext(0)
    n = 1_000_000
    keys_s, _ = make_strings(5, n)
    filt_s, _ = make_strings(6, n)
    v2 = np.random.default_rng(7).normal(0, 1, n).astype(np.float32)
    kcodes, kdict = sa.encode_col_str(keys_s)
    fcodes, fdict = sa.encode_col_str(filt_s)
    op, lo, hi = sa.str_pred_to_code(fdict, "prefix", "e")
    ngroups = len(kdict)
    fs = np.array(filt_s)
    ks = np.array(keys_s)
    mask = np.char.startswith(fs, "e")
    ecnt = np.bincount(kcodes[mask], minlength=ngroups)
    esum = np.bincount(kcodes[mask], weights=fcodes[mask].astype(np.float64),
                       minlength=ngroups).astype(np.int64)

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

    kfor = np.frombuffer(sa.encode_col_i64(kcodes), dtype=np.uint8)
    ffor = np.frombuffer(sa.encode_col_i64(fcodes), dtype=np.uint8)
    for dense in (True, False):
        if dense:
            cols = (ColView * 3)(
                ColView(kcodes.ctypes.data_as(CT.c_void_p).value, n, 0),
                ColView(fcodes.ctypes.data_as(CT.c_void_p).value, n, 0),
                ColView(v2.ctypes.data_as(CT.c_void_p).value, n, 1))
        else:
            cols = (ColView * 3)(
                ColView(kfor.ctypes.data_as(CT.c_void_p).value, n, 2),
                ColView(ffor.ctypes.data_as(CT.c_void_p).value, n, 2),
                ColView(v2.ctypes.data_as(CT.c_void_p).value, n, 1))
        tab = CT.c_void_p(0)
        rc = lib.sdb_gpu_table_load(ctx._ctx, cols, 3, CT.c_uint64(n),
                                    CT.byref(tab))
        assert rc == 0, rc
        preds = (PredSpec * 1)(PredSpec(1, op, lo, hi, 0, 0))
        aggs = (AggSpec * 2)(AggSpec(0, 0), AggSpec(1, 1))
        out = (AggResult * (ngroups * 2))()
        passed = CT.c_uint64(0)
        rc = lib.sdb_gpu_scan_agg(ctx._ctx, tab, 0, ngroups, preds, 1,
                                  aggs, 2, out, CT.byref(passed))
        assert rc == 0, rc
        assert passed.value == int(mask.sum())
        gcnt = np.array([out[g * 2 + 0].i64 for g in range(ngroups)])
        gsum = np.array([out[g * 2 + 1].i64 for g in range(ngroups)])
        np.testing.assert_array_equal(gcnt, ecnt)
        np.testing.assert_array_equal(gsum, esum)
        lib.sdb_gpu_table_free(ctx._ctx, tab)
    ctx.close()
