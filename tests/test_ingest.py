"""On-disk `.doc` ingestion tests.

No reference-built index files exist in this container (the reference's
56-submodule build is unbuildable — SURVEY.md §8c), so the fixtures are
HAND-CRAFTED files following the restated format, written by an
independent Python writer below (vints, CRC-32C, block layout, skip
levels and wand payloads re-derived from formats/posting/writer.hpp,
skip_list.hpp, format_utils.cpp — citations in sdb_host.cpp). The block
payload bytes themselves come from this repo's codec, which round 1
pinned bit-for-bit against the reference's own vendored simdcomp."""

import numpy as np
import pytest

import serenedb_amd as sa
from oracle import pyoracle as po


# ---- independent Python restatement of the container plumbing ----

def _crc32c(data):
    # CRC-32C via the reversed polynomial, bitwise (independent of the
    # table-driven C implementation)
    crc = 0xFFFFFFFF
    for b in data:
        crc ^= b
        for _ in range(8):
            crc = (crc >> 1) ^ (0x82F63B78 if crc & 1 else 0)
    return crc ^ 0xFFFFFFFF


def _vint(v):
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


FMT = b"iresearch_10_postings_documents"


def write_doc_file(terms, version=1):
    """terms: list of (docs u32 array, freqs u32 array). Returns
    (file bytes, sidecar metas). Freq feature on, no positions."""
    out = bytearray()
    out += (0x3FD76C17).to_bytes(4, "little")
    out += _vint(len(FMT)) + FMT
    out += int(version).to_bytes(4, "little")

    metas = []
    for docs, freqs in terms:
        docs = np.asarray(docs, dtype=np.uint32)
        freqs = np.asarray(freqs, dtype=np.uint32)
        df = len(docs)
        meta = {"docs_count": df, "total_freq": int(freqs.sum()),
                "doc_start": len(out), "e_single_doc": 0,
                "e_skip_start": 0}
        if df == 0:
            metas.append(meta)
            continue
        if df == 1:
            meta["e_single_doc"] = int(docs[0]) - 1
            metas.append(meta)
            continue
        doc_start = len(out)
        full = df // 128
        # skip machinery state (skip_list.hpp Skip / WriteSkip)
        skip_ptr = {}           # level -> last doc_ptr (abs)
        levels = {}             # level -> bytearray
        max_levels = 10

        def wand_entry(bi):
            lo = bi * 128
            hi = lo + 128
            f = int(freqs[lo:hi].max())
            payload = _vint(f)  # norm==freq: only freq written
            return bytes([len(payload)]) + payload

        prev = 0
        for b in range(full):
            blk = docs[b * 128:(b + 1) * 128]
            out += sa.encode_doc_block(blk, prev)
            out += sa.encode_freq_block(freqs[b * 128:(b + 1) * 128])
            prev = int(blk[-1])
            # a skip entry is written when the NEXT doc begins
            # (writer.hpp:733: count%128==0 and buffer just flushed)
            ndocs = (b + 1) * 128
            if ndocs < df:
                count = ndocs
                # level 0
                lvl = levels.setdefault(0, bytearray())
                ptr = len(out) - 0  # abs position in file
                lvl += _vint(prev)
                lvl += _vint(ptr - skip_ptr.get(0, doc_start))
                skip_ptr[0] = ptr
                lvl += wand_entry(b)
                child = len(lvl)
                count //= 128
                i = 1
                while count % 32 == 0 and i < max_levels:
                    lv = levels.setdefault(i, bytearray())
                    lv += _vint(prev)
                    lv += _vint(ptr - skip_ptr.get(i, doc_start))
                    skip_ptr[i] = ptr
                    lv += wand_entry(b)
                    nc = len(lv)
                    lv += _vint(child)
                    child = nc
                    count //= 32
                    i += 1
        has_skip = df > 128

        def root_wand():
            f = int(freqs.max())
            payload = _vint(f)
            return bytes([len(payload)]) + payload

        tail = df % 128
        if not has_skip:
            out += root_wand()
            if tail:
                out += sa.encode_doc_block(docs[full * 128:], prev)
                out += sa.encode_freq_block(freqs[full * 128:])
        else:
            if tail:
                out += sa.encode_doc_block(docs[full * 128:], prev)
                out += sa.encode_freq_block(freqs[full * 128:])
            meta["e_skip_start"] = len(out) - doc_start
            out += root_wand()
            nl = max(levels) + 1
            out += _vint(nl)
            for lv in range(nl - 1, -1, -1):
                out += _vint(len(levels[lv]))
                out += bytes(levels[lv])
        metas.append(meta)

    out += (0x3FD76C17 ^ 0xFFFFFFFF).to_bytes(4, "little")  # see below
    # kFooterMagic = -kFormatMagic (two's complement, LE)
    out[-4:] = ((-0x3FD76C17) & 0xFFFFFFFF).to_bytes(4, "little")
    out += (0).to_bytes(4, "little")
    out += int(_crc32c(bytes(out))).to_bytes(8, "little")
    return bytes(out), metas


def synth_terms(seed, doc_count, sels):
    return [sa.synth_postings(seed, doc_count, t, s)
            for t, s in enumerate(sels)]


def test_ingest_roundtrip_small():
    """Hand-crafted .doc -> ingest -> MUST byte-equal the segment built
    directly from the same postings (same builder), and execute
    identically through the oracle. Covers df==1, df==128 (block, no
    skip), df in (1,128), df>128 (skip list), df>4096 (level-1 skips)."""
    doc_count = 60_000
    rng = np.random.default_rng(7)
    terms = synth_terms(7, doc_count, [0.10, 0.002, 0.0001])
    # plant exact-df edge cases
    d1 = np.sort(rng.choice(np.arange(1, doc_count + 1), 1,
                            replace=False)).astype(np.uint32)
    d128 = np.sort(rng.choice(np.arange(1, doc_count + 1), 128,
                              replace=False)).astype(np.uint32)
    d129 = np.sort(rng.choice(np.arange(1, doc_count + 1), 129,
                              replace=False)).astype(np.uint32)
    for d in (d1, d128, d129):
        terms.append((d, rng.integers(1, 9, len(d)).astype(np.uint32)))
    fbytes, metas = write_doc_file(terms)
    norms = sa.synth_norms(7, doc_count)
    blob = sa.ingest_doc(fbytes, metas, doc_count, norms)
    ref = sa.build_segment(doc_count, terms, norms)
    assert blob == ref, "ingested segment must byte-equal the direct build"
    hits, total = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 50)
    hits2, total2 = po.execute_topk([ref], [0, 1, 2], [1.0] * 3, 50)
    assert total == total2
    np.testing.assert_array_equal(hits["doc"], hits2["doc"])


def test_ingest_multilevel_skip():
    """df > 4096 exercises level-1 skip entries (every 32 level-0
    entries, skip_list.hpp Skip cadence)."""
    doc_count = 40_000
    terms = synth_terms(9, doc_count, [0.30])
    assert len(terms[0][0]) > 4096 * 2
    fbytes, metas = write_doc_file(terms)
    blob = sa.ingest_doc(fbytes, metas, doc_count)
    ref = sa.build_segment(doc_count, terms)
    assert blob == ref


def test_ingest_rejects_corruption():
    doc_count = 20_000
    terms = synth_terms(11, doc_count, [0.05, 0.01])
    fbytes, metas = write_doc_file(terms)
    good = bytearray(fbytes)

    def expect_fail(mut, metas=metas):
        with pytest.raises(ValueError):
            sa.ingest_doc(bytes(mut), metas, doc_count)

    b = bytearray(good); b[0] ^= 0xFF; expect_fail(b)       # header magic
    b = bytearray(good); b[10] ^= 0x01; expect_fail(b)      # format name
    b = bytearray(good); b[-20] ^= 0x01; expect_fail(b)     # checksum
    b = bytearray(good); b[-16] ^= 0xFF; expect_fail(b)     # footer magic
    expect_fail(good[:len(good) // 2])                      # truncation
    # a flipped payload byte inside a block: CRC catches it
    b = bytearray(good); b[60] ^= 0x40; expect_fail(b)
    # bad sidecar: doc_start past the file
    m2 = [dict(m) for m in metas]; m2[0]["doc_start"] = len(good) + 5
    expect_fail(good, m2)
    # bad sidecar: wrong skip offset (cross-check -80)
    m3 = [dict(m) for m in metas]
    m3[0]["e_skip_start"] = m3[0]["e_skip_start"] + 1
    expect_fail(good, m3)
    # pristine still works
    sa.ingest_doc(bytes(good), metas, doc_count)


@pytest.mark.gpu
def test_ingest_gpu_execute():
    """Ingested reference-format postings execute on the GPU identically
    to the oracle."""
    doc_count = 300_000
    terms = synth_terms(13, doc_count, [0.08, 0.03, 0.01])
    fbytes, metas = write_doc_file(terms)
    norms = sa.synth_norms(13, doc_count)
    blob = sa.ingest_doc(fbytes, metas, doc_count, norms)
    ctx = sa.GpuContext(0)
    seg = ctx.load_segment(blob)
    hits, total = ctx.execute_topk([seg], [0, 1, 2], [1.0] * 3, 500)
    ohits, ototal = po.execute_topk([blob], [0, 1, 2], [1.0] * 3, 500)
    assert total == ototal
    np.testing.assert_array_equal(hits["doc"], ohits["doc"])
    np.testing.assert_array_equal(hits["score"].view(np.uint32),
                                  ohits["score"].view(np.uint32))
    ctx.close()
