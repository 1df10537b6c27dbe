/* sdb_format.h — SereneDB-AMD segment format v1 (shared by host C++, HIP
 * kernels, and the CPU oracle).
 *
 * The CONTAINER (header/term table/block descriptors/norms layout) is this
 * project's own MI355X-first design: a flat, random-access block-descriptor
 * table replaces the reference's sequential skip lists
 * (libs/iresearch/include/iresearch/formats/posting/skip_list.hpp:42-255),
 * because GPU workgroups address blocks by doc-window, not by forward seeks.
 * Each descriptor carries `prev_doc` (the last doc-id of the previous block)
 * so any 128-doc block decodes independently of its predecessors — the
 * delta-decode base the reference carries implicitly while streaming
 * (formats/posting/iterator_doc.hpp:36-344).
 *
 * The BLOCK PAYLOAD ENCODINGS are bit-for-bit the reference's FormatTraits128
 * ("1_5simd") families:
 *   - doc blocks  : formats/posting/format_block_128.hpp:51-242 (write) /
 *                   :446-559 (read)  — DeltaEncoding enum :652-712
 *   - freq blocks : format_block_128.hpp:244-379 (write) / :561-636 (read)
 *                   — Encoding enum :722-771
 *   - delta bitpack layout = simdcomp simdpackwithoutmaskd1/simdunpackd1
 *     (third_party/simdcomp/src/simdintegratedbitpacking.c): plain sequential
 *     deltas (d1), packed "vertically": value index i -> SSE lane c = i%4,
 *     group g = i/4; lane c's 32 deltas form a little-endian bitstream of
 *     32*b bits packed LSB-first into b 32-bit words; the 4 lanes' word
 *     streams are interleaved at 32-bit-word granularity
 *     (flat_word[w*4 + c] = word w of lane c's stream).
 *   - non-delta bitpack = simdpackwithoutmask/simdunpack
 *     (third_party/simdcomp/src/simdbitpacking.c), same vertical layout,
 *     no delta/prefix-sum.
 *   - streamvbyte / streamvbyte_delta = the public Lemire streamvbyte
 *     "1234" format (1/2/3/4-byte lanes; 2-bit codes, low bits first, all
 *     control bytes precede all data bytes). The reference's streamvbyte
 *     submodule is EMPTY (URL only in .gitmodules, no pinned SHA), so this
 *     restates the published format; parity is anchored at the call sites
 *     format_block_128.hpp:182-199,507-517 and at round-trip tests.
 *
 * Doc-id conventions follow the reference (utils/type_limits.hpp:41-51):
 * invalid = 0, min = 1, eof = 0xFFFFFFFF, block size = 128.
 */
#ifndef SDB_FORMAT_H
#define SDB_FORMAT_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define SDB_BLOCK_SIZE 128u
#define SDB_DOC_INVALID 0u
#define SDB_DOC_MIN 1u
#define SDB_DOC_EOF 0xFFFFFFFFu

/* DeltaEncoding — doc blocks (format_block_128.hpp:652-712). Values match the
 * reference byte tags exactly (payload byte 0 of every block). */
enum SdbDeltaEncoding {
  SDB_DE_VALUES = 0,
  SDB_DE_DELTA_ALL_SAME_08 = 1,
  SDB_DE_DELTA_ALL_SAME_16 = 2,
  SDB_DE_DELTA_ALL_SAME_32 = 3,
  SDB_DE_FOR_BITSET = 4,
  SDB_DE_STREAMVBYTE1234 = 5,
  SDB_DE_FOR_STREAMVBYTE1234 = 6, /* reserved, never written (reference too) */
  SDB_DE_DELTA_STREAMVBYTE1234 = 7,
  SDB_DE_DELTA_BITPACK_02 = 8, /* .. SDB_DE_DELTA_BITPACK_31 = 37 */
};
#define SDB_DE_DELTA_BITPACK(bits) (SDB_DE_DELTA_BITPACK_02 + ((bits)-2))

/* Encoding — freq blocks (format_block_128.hpp:722-771). */
enum SdbEncoding {
  SDB_E_VALUES = 0,
  SDB_E_ALL_SAME_08 = 1,
  SDB_E_ALL_SAME_16 = 2,
  SDB_E_ALL_SAME_32 = 3,
  SDB_E_STREAMVBYTE1234 = 4,
  SDB_E_BITPACK_01 = 5, /* .. SDB_E_BITPACK_31 = 35 */
};
#define SDB_E_BITPACK(bits) (SDB_E_BITPACK_01 + ((bits)-1))

/* One 128-doc postings block. Replaces a level-0 skip entry
 * (skip_list.hpp:42-255: {last doc, doc_ptr, wand payload}) with a flat
 * random-access record. Offsets are relative to the owning term's payload
 * span. max_freq/min_norm are an admissible WAND-style score-bound pair
 * (upper-bounds any monotone BM25 score in the block; cf. FreqNormWriter
 * tags, index/norm.hpp:108-132) — unused by the exact two-pass round-1 path.
 */
typedef struct SdbBlockDesc {
  uint32_t prev_doc; /* last doc of previous block; base for delta decode */
  uint32_t last_doc; /* last doc of this block */
  uint32_t doc_off;  /* byte offset of doc-block payload (incl. tag byte) */
  uint32_t freq_off; /* byte offset of freq-block payload (incl. tag byte) */
  uint16_t len;      /* docs in this block: 128, or 1..127 for the tail */
  uint16_t flags;    /* v2: bit0=1 -> the fused shape (128 docs, all three
                      * streams bitpacked) and bits1-5/6-10/11-15 carry the
                      * doc/freq/norm bit widths, so the decode issues every
                      * packed-word load without first fetching the payload
                      * tag bytes (one full memory round trip per block);
                      * bit0=0 -> bits1+ = freq-block byte size (norm_off =
                      * freq_off + (flags>>1)) */
  uint32_t max_freq; /* max freq in block  (WAND bound, round-2 pruning) */
  uint32_t min_norm; /* min norm over block's docs (WAND bound) */
} SdbBlockDesc;

typedef struct SdbTermEntry {
  uint64_t desc_begin;    /* first SdbBlockDesc index */
  uint64_t desc_end;      /* one past last */
  uint64_t payload_begin; /* byte offset into segment payload section */
  uint64_t payload_end;
  uint32_t df;            /* docs_with_term (postings count) */
  uint32_t max_freq;      /* term-level max freq */
  uint64_t total_freq;    /* sum of freqs (unused by BM25; kept for parity) */
} SdbTermEntry;

#define SDB_SEG_MAGIC 0x3130444D41424453ull /* "SDBAMD01" */

/* Serialized segment header. All section offsets are bytes from the start of
 * the blob; every section start is 64-byte aligned. The whole blob is the
 * "mmap-equivalent byte span" handed across the C ABI (cf. the reference's
 * mmap directory zero-copy reads, store/data_input.hpp:59,99). */
typedef struct SdbSegHeader {
  uint64_t magic;
  uint32_t version;
  uint32_t nterms;
  uint32_t doc_count;       /* docs are 1..doc_count */
  uint32_t docs_with_field; /* BM25 field stat (== doc_count here) */
  uint64_t total_term_freq; /* BM25 field stat: sum of norms */
  uint64_t total_blocks;
  uint64_t off_terms;   /* SdbTermEntry[nterms] */
  uint64_t off_desc;    /* SdbBlockDesc[total_blocks] */
  uint64_t off_norms;   /* uint32_t[doc_count+1]; norms[0] unused (doc 0 invalid) */
  uint64_t off_payload; /* byte stream */
  uint64_t payload_size;
  uint64_t blob_size;
} SdbSegHeader;

/* Term metadata sidecar for on-disk `.doc` ingestion — exactly the
 * TermMetaImpl fields the reference's burst-trie terms dict serializes per
 * term (formats_attributes.hpp:30; PostingsWriterBase::Encode,
 * formats/posting/writer.hpp:379-412). The burst-trie reader itself is out
 * of scope this round: callers supply these from a sidecar.
 */
typedef struct SdbDocTermMeta {
  uint32_t docs_count;   /* df */
  uint64_t total_freq;   /* sum of freqs (meta.freq) */
  uint64_t doc_start;    /* absolute offset of the term's stream in .doc */
  uint32_t e_single_doc; /* df==1: doc - doc_limits::min() */
  uint64_t e_skip_start; /* df>128: skip area offset relative to doc_start */
} SdbDocTermMeta;

/* Parsed (in-memory) view over a segment blob. */
typedef struct SdbSegmentView {
  const SdbSegHeader* hdr;
  const SdbTermEntry* terms;
  const SdbBlockDesc* desc;
  const uint32_t* norms; /* indexed by doc id (1-based; [0] unused) */
  const uint8_t* payload;
} SdbSegmentView;

#ifdef __cplusplus
}
#endif
#endif /* SDB_FORMAT_H */
