"""WAND block-max pruning demo (the shape Block-Max WAND exists for).

Skewed term frequencies: ~1 posting in 1009 carries freq 200, the rest
freq 1. The k-th threshold locks onto the spike scores; every postings
block whose descriptor bound (max_freq, min_norm) falls below it is
skipped without decoding. On doc-uniform rare terms block-max CANNOT
prune (a sparse term's blocks span huge doc ranges, so every window sees
the block's full upper bound) — that null result is documented in
tools/ROUND2_NOTES.md.

Prints ms/query and visited-vs-total matches for wand on/off and checks
the hit sets match bit-for-bit (WAND is exact).
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import serenedb_amd as sa

doc_count = 100_000_000
k = 10

# single term, sel-0.25 hashed membership -> delta-bitpack blocks (real
# decode work; arange strides encode as all-same deltas, nearly free to
# decode). Measured shape notes (tools/ROUND2_NOTES.md): doc-uniform rare
# terms cannot prune (their blocks span huge doc ranges); N-term plans
# where EVERY term carries spikes cannot prune either (the sum bound
# own + sum(other wub) always clears tau) -- freq skew within a term is
# the shape block-max bounds act on, as in the BMW literature.
docs, _ = sa.synth_postings(79, doc_count, 0, 0.25)  # ~25M postings
freqs = np.ones(len(docs), dtype=np.uint32)
freqs[::1009] = 200  # spike ~1 posting per 8 blocks
norms = sa.synth_norms(79, doc_count)
blob = sa.build_segment(doc_count, [(docs, freqs)], norms)
NT = 1
ctx = sa.GpuContext(0)
seg = ctx.load_segment(blob)


def run(wand, reps=20):
    best = 1e9
    hits = total = None
    for _ in range(reps):
        t0 = time.time()
        hits, total = ctx.execute_topk([seg], list(range(NT)),
                                       [1.0] * NT, k, wand=wand)
        best = min(best, time.time() - t0)
    return hits, total, best


base, total, t_off = run(False)
wnd, visited, t_on = run(True)
np.testing.assert_array_equal(base["doc"], wnd["doc"])
np.testing.assert_array_equal(base["score"].view(np.uint32),
                              wnd["score"].view(np.uint32))
print(f"wand=off: {t_off*1e3:.3f} ms/query, total_matches={total}")
print(f"wand=on : {t_on*1e3:.3f} ms/query, visited={visited} "
      f"({100.0*visited/total:.1f}% of {total}), exact top-{k} identical; "
      f"speedup {t_off/t_on:.2f}x")
