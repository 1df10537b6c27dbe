"""Block-codec parity tests.

Pins:
  1. The delta-bitpack / bitpack bit layout against the REFERENCE'S OWN
     vendored simdcomp (compiled unmodified from
     /root/reference/third_party/simdcomp into oracle/_ref) — the exact
     arithmetic the reference executes in
     formats/posting/format_block_128.hpp:235,372,553,630.
  2. Product encoder bytes == oracle encoder bytes (independent restatements
     of WriteTailDelta/WriteTail must agree byte-for-byte).
  3. Round-trip decode(encode(x)) == x through BOTH decoders, for every
     encoding family and every tail length (mirrors the reference's
     formats_15_tests.cpp round-trip strategy: N(mu,sigma) doc gaps,
     Singleton/Short/Block/Medium/Long shapes).
"""

import numpy as np
import pytest

import serenedb_amd as sa
from oracle import pyoracle as po


def gen_docs(rng, n, gap_dist="normal", mu=8.0, sigma=3.0, prev=0):
    """sorted unique doc ids with N(mu,sigma) gaps (formats_15_tests shape)"""
    if gap_dist == "normal":
        gaps = np.maximum(1, rng.normal(mu, sigma, n).astype(np.int64))
    elif gap_dist == "one":
        gaps = np.ones(n, dtype=np.int64)
    elif gap_dist == "large":
        gaps = rng.integers(1, 1 << 24, n)
    else:
        gaps = rng.integers(1, 64, n)
    docs = prev + np.cumsum(gaps)
    return docs.astype(np.uint32)


# ---------------------------------------------------------------------------
# 1. bit layout vs the reference's own simdcomp
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("bits", range(2, 32))
def test_delta_bitpack_layout_vs_reference_simdcomp(bits):
    if po.ref_simdcomp() is None:
        pytest.skip("reference simdcomp not built (no /root/reference)")
    rng = np.random.default_rng(1000 + bits)
    prev = np.uint32(rng.integers(0, 1 << 20))
    maxd = (1 << bits) - 1
    gaps = rng.integers(1, maxd + 1, 128).astype(np.uint64)
    docs = (prev + np.cumsum(gaps)).astype(np.uint32)
    ref_bytes = po.ref_pack_d1(int(prev), docs, bits)
    # my encoders emit tag byte first; compare payloads
    mine = sa.encode_doc_block(docs, int(prev))
    # ensure the encoder actually chose this bitpack width
    if mine[0] != 8 + bits - 2:
        pytest.skip(f"encoder chose family {mine[0]} (sizes degenerate)")
    assert mine[1:] == ref_bytes
    # and the oracle decode of the REFERENCE bytes reproduces the docs
    dec = po.ref_unpack_d1(int(prev), ref_bytes, bits)
    np.testing.assert_array_equal(dec, docs)
    odec, _ = po.decode_doc_block(bytes([mine[0]]) + ref_bytes, 128,
                                  int(prev))
    np.testing.assert_array_equal(odec, docs)


@pytest.mark.parametrize("bits", range(1, 32))
def test_bitpack_layout_vs_reference_simdcomp(bits):
    if po.ref_simdcomp() is None:
        pytest.skip("reference simdcomp not built")
    rng = np.random.default_rng(2000 + bits)
    vals = rng.integers(0, 1 << bits, 128).astype(np.uint32)
    vals[0] = (1 << bits) - 1  # defeat all_same and force max width
    if bits > 1:
        vals[1] = 0
    ref_bytes = po.ref_pack(vals, bits)
    mine = sa.encode_freq_block(vals)
    if mine[0] != 5 + bits - 1:
        pytest.skip(f"encoder chose family {mine[0]}")
    assert mine[1:] == ref_bytes
    dec = po.ref_unpack(ref_bytes, bits)
    np.testing.assert_array_equal(dec, vals)


# ---------------------------------------------------------------------------
# 2. product encoder == oracle encoder, both decoders agree
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("seed", range(20))
def test_doc_block_roundtrip_random(seed):
    rng = np.random.default_rng(seed)
    n = int(rng.integers(1, 129))
    kind = ["normal", "one", "large", "small"][seed % 4]
    prev = int(rng.integers(0, 1 << 16))
    docs = gen_docs(rng, n, kind, prev=prev)
    enc_h = sa.encode_doc_block(docs, prev)
    enc_o = po.encode_doc_block(docs, prev)
    assert enc_h == enc_o, f"family h={enc_h[0]} o={enc_o[0]}"
    dec_h, con_h = sa.decode_doc_block(enc_h, n, prev)
    dec_o, con_o = po.decode_doc_block(enc_h, n, prev)
    assert con_h == con_o == len(enc_h)
    np.testing.assert_array_equal(dec_h, docs)
    np.testing.assert_array_equal(dec_o, docs)


@pytest.mark.parametrize("seed", range(20))
def test_freq_block_roundtrip_random(seed):
    rng = np.random.default_rng(100 + seed)
    n = int(rng.integers(1, 129))
    choice = seed % 5
    if choice == 0:
        freqs = np.full(n, int(rng.integers(0, 255)), dtype=np.uint32)
    elif choice == 1:
        freqs = np.full(n, int(rng.integers(256, 1 << 16)), dtype=np.uint32)
    elif choice == 2:
        freqs = rng.integers(1, 8, n).astype(np.uint32)
    elif choice == 3:
        freqs = rng.integers(0, 1 << 20, n).astype(np.uint32)
    else:
        freqs = rng.integers(0, 1 << 31, n).astype(np.uint32)
    enc_h = sa.encode_freq_block(freqs)
    enc_o = po.encode_freq_block(freqs)
    assert enc_h == enc_o
    dec_h, con_h = sa.decode_freq_block(enc_h, n)
    dec_o, con_o = po.decode_freq_block(enc_h, n)
    assert con_h == con_o == len(enc_h)
    np.testing.assert_array_equal(dec_h, freqs)
    np.testing.assert_array_equal(dec_o, freqs)


# ---------------------------------------------------------------------------
# 3. every family explicitly
# ---------------------------------------------------------------------------
def _roundtrip_doc(docs, prev, want_family=None):
    docs = np.asarray(docs, dtype=np.uint32)
    enc = sa.encode_doc_block(docs, prev)
    if want_family is not None:
        assert enc[0] == want_family, f"got family {enc[0]}"
    assert enc == po.encode_doc_block(docs, prev)
    dec, _ = po.decode_doc_block(enc, len(docs), prev)
    np.testing.assert_array_equal(dec, docs)
    dec2, _ = sa.decode_doc_block(enc, len(docs), prev)
    np.testing.assert_array_equal(dec2, docs)


def test_family_all_same():
    _roundtrip_doc(np.arange(1, 129) * 3, 0, want_family=1)  # delta=3 u8
    _roundtrip_doc(np.arange(1, 129) * 300, 0, want_family=2)  # u16
    _roundtrip_doc(np.arange(1, 129, dtype=np.uint64) * 70000, 0,
                   want_family=3)  # u32


def test_family_bitset():
    # dense block: 128 docs in a span of ~160 -> bitset beats 2-bit bitpack?
    # 2-bit pack = 32B; bitset span 192 -> 3 words = 25B -> chosen.
    rng = np.random.default_rng(7)
    docs = np.sort(rng.choice(np.arange(1, 180), 128, replace=False))
    _roundtrip_doc(docs.astype(np.uint32), 0, want_family=4)


def test_family_tail_streamvbyte():
    rng = np.random.default_rng(8)
    # tail with large absolute values but huge deltas -> svb over values
    docs = np.sort(rng.choice(np.arange(1, 200), 40, replace=False))
    prev = 0
    enc = sa.encode_doc_block(docs.astype(np.uint32), prev)
    assert enc[0] in (4, 5, 7)  # bitset or svb families for tails
    _roundtrip_doc(docs.astype(np.uint32), prev)


def test_family_tail_delta_svb():
    # sparse tail, large doc ids: delta svb much smaller than values/svb
    docs = (1 << 25) + np.cumsum(
        np.random.default_rng(9).integers(1, 200, 50)).astype(np.uint32)
    enc = sa.encode_doc_block(docs, 0)
    assert enc[0] == 7, enc[0]
    _roundtrip_doc(docs, 0)


def test_family_values_fallback():
    # deltas needing >=32 bits force raw values
    docs = np.array([1, 0x80000002, 0xFFFFFFFE], dtype=np.uint32)
    enc = sa.encode_doc_block(docs, 0)
    _roundtrip_doc(docs, 0)


def test_tail_every_length():
    rng = np.random.default_rng(11)
    for n in range(1, 128):
        prev = int(rng.integers(0, 1000))
        docs = gen_docs(rng, n, "normal", prev=prev)
        _roundtrip_doc(docs, prev)
        freqs = rng.integers(1, 256, n).astype(np.uint32)
        enc = sa.encode_freq_block(freqs)
        assert enc == po.encode_freq_block(freqs)
        dec, _ = po.decode_freq_block(enc, n)
        np.testing.assert_array_equal(dec, freqs)


def test_block_every_bitwidth_roundtrip():
    rng = np.random.default_rng(12)
    for bits in range(2, 32):
        maxd = (1 << bits) - 1
        gaps = rng.integers(max(1, maxd // 2), maxd + 1, 128)
        gaps[0] = maxd  # pin the width
        gaps[1] = 1
        docs = np.cumsum(gaps).astype(np.uint32)
        if int(docs[-1]) >= 0xFFFFFFFF:
            continue
        _roundtrip_doc(docs, 0)


def test_v2_norm_stream_matches_column():
    """v2 segments embed per-block norm streams (flags carries either the
    packed bit widths for the fused shape or the freq-block size —
    sdb_format.h);
    decoded values must equal the norm column entries for the block's docs —
    the index-build-time materialization of the reference's norm-column
    gather."""
    import ctypes as CT

    rng = np.random.default_rng(33)
    doc_count = 5000
    docs = np.sort(rng.choice(np.arange(1, doc_count + 1, dtype=np.uint32),
                              700, replace=False))
    freqs = rng.integers(1, 200, len(docs)).astype(np.uint32)
    norms = rng.integers(1, 3000, doc_count + 1).astype(np.uint32)
    blob = sa.build_segment(doc_count, [(docs, freqs)], norms)
    buf = np.frombuffer(blob, dtype=np.uint8)
    hdr = np.frombuffer(blob[:16], dtype=np.uint32)
    assert hdr[2] == 3, "expect format v3"

    class _View(CT.Structure):
        _fields_ = [("hdr", CT.c_void_p), ("terms", CT.c_void_p),
                    ("desc", CT.c_void_p), ("norms", CT.c_void_p),
                    ("payload", CT.c_void_p)]

    v = _View()
    rc = sa.host().sdb_host_segment_parse(
        buf.ctypes.data_as(CT.c_void_p), CT.c_uint64(len(buf)), CT.byref(v))
    assert rc == 0
    # walk descriptors: decode the norm block after each freq block
    import struct
    hdr_full = struct.unpack("<QIIIIQQQQQQQQ", blob[:88])
    off_terms, off_desc, off_norms, off_payload = hdr_full[7:11]
    nblocks = hdr_full[6]
    pos = 0
    for b in range(nblocks):
        d = struct.unpack("<IIIIHHII", blob[off_desc + 28 * b:
                                            off_desc + 28 * b + 28])
        prev, last, doc_off, freq_off, length, flags = d[:6]
        if flags & 1:  # fused: freq-block size = 1 + 16*fbits
            fsize = 1 + 16 * ((flags >> 6) & 31)
            # the carried widths must match the payload tags
            assert blob[off_payload + doc_off] - 8 + 2 == (flags >> 1) & 31
            assert blob[off_payload + freq_off] - 5 + 1 == (flags >> 6) & 31
            assert (blob[off_payload + freq_off + fsize] - 5 + 1 ==
                    (flags >> 11) & 31)
        else:
            fsize = flags >> 1
        norm_payload = blob[off_payload + freq_off + fsize:
                            off_payload + freq_off + fsize + 600]
        dec, _ = po.decode_freq_block(norm_payload, length)
        np.testing.assert_array_equal(dec, norms[docs[pos:pos + length]])
        pos += length


def test_for_column_codec_roundtrip():
    """FoR/bitpack i64 column codec: lossless round trip + compression"""
    rng = np.random.default_rng(77)
    for vals in (
        rng.integers(0, 1 << 20, 200_000).astype(np.int64),
        rng.integers(-500, 500, 100_000).astype(np.int64),
        np.full(70_000, 42, dtype=np.int64),          # width 0
        np.sort(rng.integers(0, 1 << 31, 130_000)).astype(np.int64),
    ):
        blob = sa.encode_col_i64(vals, group_rows=65536)
        dec = sa.decode_col_i64(blob, len(vals))
        np.testing.assert_array_equal(dec, vals)
    # 20-bit values compress to ~2.5B/row
    v = rng.integers(0, 1 << 20, 1_000_000).astype(np.int64)
    blob = sa.encode_col_i64(v)
    assert len(blob) < 2.7 * len(v), len(blob) / len(v)


def test_segment_blob_format_stability():
    """Golden hash of a small synthetic segment blob: the serialized format
    (container layout + reference-exact block encodings + v2 norm streams)
    must not drift silently across commits. Update CONSCIOUSLY on any
    deliberate format change (bump SdbSegHeader.version too)."""
    import hashlib

    blob = sa.build_synth_segment(123, 1, 50_000, [0.07, 0.02])
    h = hashlib.sha256(blob).hexdigest()
    assert h == GOLDEN_BLOB_SHA, (
        f"segment format changed: {h} (deliberate? update GOLDEN_BLOB_SHA "
        "and bump the format version)")


# v3 (round 2): descriptor flags carry the fused-shape bit + packed bit
# widths (sdb_format.h) — deliberate container change, version bumped 2->3;
# the PAYLOAD block encodings (the reference-pinned bytes) are unchanged
# and separately pinned by the codec round-trip + simdcomp tests above.
GOLDEN_BLOB_SHA = "15b95522b8b8eb2146c1609af94acef00ceaf542e02dcadb8eea8de819ea4fe7"
