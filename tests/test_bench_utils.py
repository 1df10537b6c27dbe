"""Unit tests for bench.py's distributed merge helpers (pack_hits /
unpack_hits): the packed int64 keys must sort identically to the
(score desc, doc asc) total order the engine reports, including exact
score ties and doc-id offsets for sharded ranks."""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench


def _hits(scores, docs):
    dt = np.dtype([("score", "f4"), ("doc", "u4"), ("segment", "u4")])
    h = np.zeros(len(scores), dtype=dt)
    h["score"] = scores
    h["doc"] = docs
    return h


def test_pack_roundtrip():
    scores = np.array([3.5, 1.25, 0.001], dtype=np.float32)
    docs = np.array([7, 100, 4_000_000_000], dtype=np.uint32)
    packed = bench.pack_hits(_hits(scores, docs), base=0, k=3)
    s2, d2 = bench.unpack_hits(packed)
    np.testing.assert_array_equal(s2.view(np.uint32),
                                  scores.view(np.uint32))
    np.testing.assert_array_equal(d2, docs)


def test_pack_sort_order_matches_engine_order():
    # equal scores must tie-break doc ASC after a descending sort of the
    # packed keys (the doc field is stored inverted)
    scores = np.array([2.0, 2.0, 5.0, 2.0, 0.5], dtype=np.float32)
    docs = np.array([50, 3, 9, 17, 1], dtype=np.uint32)
    packed = bench.pack_hits(_hits(scores, docs), base=0, k=5)
    packed.sort()
    s2, d2 = bench.unpack_hits(packed[::-1])
    np.testing.assert_array_equal(s2, [5.0, 2.0, 2.0, 2.0, 0.5])
    np.testing.assert_array_equal(d2, [9, 3, 17, 50, 1])


def test_pack_base_offset_and_padding():
    # sharded ranks add their doc base; k > len(hits) pads with minimal
    # keys that sort below every real hit
    scores = np.array([1.0], dtype=np.float32)
    docs = np.array([5], dtype=np.uint32)
    packed = bench.pack_hits(_hits(scores, docs), base=1000, k=3)
    assert len(packed) == 3
    packed_sorted = np.sort(packed)[::-1]
    s2, d2 = bench.unpack_hits(packed_sorted[:1])
    assert s2[0] == 1.0 and d2[0] == 1005
    assert (packed_sorted[1:] <= packed_sorted[0]).all()
