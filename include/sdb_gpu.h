/* sdb_gpu.h — C ABI of the MI355X-native SereneDB search/analytics hot path.
 *
 * This is the drop-in boundary of SURVEY.md §8(b): the entry points a
 * maintainer of the reference would bind where its scan operator hands a
 * prepared per-segment query to the execution hot loop:
 *
 *   sdb_gpu_execute_topk  replaces the per-segment collect loop
 *     CollectSegmentTopK / RunTopKScan
 *     (server/connector/duckdb_search_full_scan.cpp:1868-1943) and the
 *     standalone driver irs::ExecuteTopK/ExecuteTopKWithCount
 *     (libs/iresearch/include/iresearch/search/doc_collector.hpp:44-136),
 *     i.e. DocIterator::Collect (index/iterators.hpp:297-298,400-428) over a
 *     BlockDisjunction/Conjunction (search/block_disjunction.hpp:122-732,
 *     search/conjunction.hpp:248-529) scored by BM25
 *     (search/bm25.cpp:60-109,279-310) into an NthPartitionScoreCollector
 *     (index/iterators.hpp:103-253).
 *
 *   sdb_gpu_scan_agg  replaces FullScanner::Scan + ColFilterChain predicate
 *     narrowing (server/connector/full_scanner.h:40-90,
 *     index/table_filter_iterator.hpp:104-312) feeding the external DuckDB
 *     PhysicalHashAggregate (un-vendored; parity pinned at SQL-result level,
 *     SURVEY.md §8c).
 *
 * Conventions: every function returns 0 on success or a negative SDB_ERR_*
 * code; all buffers are caller-allocated; an opaque context owns the device,
 * streams and workspace; one context is single-threaded, use one per thread.
 * Hot entry points never allocate on the device after segment load. The
 * implementation REQUIRES a GPU: calls fail loudly (SDB_ERR_NO_GPU) rather
 * than fall back to any CPU path.
 *
 * ScoreDoc mirrors irs::ScoreDoc (index/iterators.hpp:93-99). Doc-id
 * conventions follow utils/type_limits.hpp:41-47 (invalid=0, min=1,
 * eof=0xFFFFFFFF).
 */
#ifndef SDB_GPU_H
#define SDB_GPU_H

#include <stddef.h>
#include <stdint.h>

#include "sdb_format.h"

#ifdef __cplusplus
extern "C" {
#endif

#define SDB_OK 0
#define SDB_ERR_INVALID (-1)
#define SDB_ERR_NO_GPU (-2)
#define SDB_ERR_HIP (-3)
#define SDB_ERR_OOM (-4)
#define SDB_ERR_BAD_SEGMENT (-5)

/* mirrors irs::ScoreDoc: {score_t score; doc_id_t doc; uint32_t segment_idx}
 * (index/iterators.hpp:93-99) */
typedef struct SdbScoreDoc {
  float score;
  uint32_t doc;
  uint32_t segment_idx;
} SdbScoreDoc;

typedef struct SdbGpuCtx SdbGpuCtx;     /* opaque: device, stream, workspace */
typedef struct SdbGpuSegment SdbGpuSegment; /* opaque: device-resident segment */

/* ---- context ---- */
int sdb_gpu_ctx_create(int device, SdbGpuCtx** out);
int sdb_gpu_ctx_destroy(SdbGpuCtx* ctx);
/* version/build string (static) */
const char* sdb_gpu_version(void);

/* ---- segment residency ----
 * Uploads a segment blob (the byte span of sdb_format.h) into HBM and keeps
 * it resident — the analogue of the reference's mmap'd segment files
 * (store/mmap_directory.hpp) with 288 GB HBM3E standing in for the page
 * cache. `blob` must stay valid during the call only. */
int sdb_gpu_segment_load(SdbGpuCtx* ctx, const void* blob, size_t blob_size,
                         SdbGpuSegment** out);
int sdb_gpu_segment_free(SdbGpuCtx* ctx, SdbGpuSegment* seg);

/* ---- query plan ----
 * Terms are addressed by index into the segment's term table: the term-dict
 * lookup (burst-trie .tm) is a per-query O(#terms) CPU step in the reference
 * and stays on the host (SURVEY.md §2 "formats/index": out of scope).
 * Per-term boost mirrors irs::Filter boost. min_match = 1 is an OR
 * (BlockDisjunction), min_match = nterms an AND (Conjunction), anything
 * between mirrors BlockDisjunction's min-match counting
 * (block_disjunction.hpp:122-732 match counts). */
typedef struct SdbTermRef {
  uint32_t term_idx;
  float boost;
} SdbTermRef;

typedef enum SdbScorerType {
  SDB_SCORER_BM25 = 0,       /* search/bm25.cpp:89-109 */
  SDB_SCORER_TFIDF = 1,      /* search/tfidf.cpp:60-62: idf*sqrt(freq) */
  SDB_SCORER_TFIDF_NORM = 2, /* tfidf.cpp:72-76: idf*sqrt(freq)/sqrt(norm) */
} SdbScorerType;

typedef struct SdbQueryPlan {
  const SdbTermRef* terms;
  uint32_t nterms;
  uint32_t min_match;
  float k1; /* BM25 k (search/bm25.hpp:62: default 1.2) */
  float b;  /* BM25 b (search/bm25.hpp:64: default 0.75) */
  uint32_t scorer; /* SdbScorerType; TFIDF idf = log1p((N+1)/(df+1)),
                      tfidf.cpp:148-151 */
  /* WAND block-max pruning (max_score_iterator.hpp Block-Max MaxScore
   * analogue over the descriptor max_freq/min_norm bounds): skips postings
   * blocks that provably cannot reach the current k-th score. EXACT top-k
   * (every skipped doc's full score < the threshold); only valid for pure
   * disjunctions (min_match == 1); total_matches then counts only VISITED
   * matches (the reference's WAND path likewise stops counting,
   * reader.hpp:517-524 wand gate). 0 = off. */
  uint32_t wand;
  /* Optional GLOBAL BM25 stats for sharded (multi-rank) execution, where
   * each device holds only its shard: the cross-rank stats merge is the
   * PreparePhase barrier analogue (duckdb_search_full_scan.cpp:1359-1384)
   * done over RCCL by the caller. Zero/NULL = derive from the segments
   * passed to this call (single-node ExecuteTopKWithCount semantics). */
  uint64_t g_docs_with_field;
  uint64_t g_total_term_freq;
  const uint64_t* g_docs_with_term; /* per plan term, or NULL */
  /* HasFilterBoost scorer variants (bm25.cpp:112-140 Bm1Boost, :69-109
   * Bm15/Bm25 boost[i]*num; tfidf equivalents): when nonzero, every
   * segment must have a boost column attached (sdb_gpu_segment_attach_
   * boost) and each term contribution is multiplied by boost[doc]. With
   * k1 == 0 this is BM1's only nonzero form: score = boost[doc]*num.
   * (The reference's Bm1Boost ASSIGNS rather than merges under multi-term
   * disjunctions; this implementation sums term contributions uniformly —
   * single-term BM1 parity is exact, the multi-term BM1 merge order is a
   * documented refinement.) */
  uint32_t filter_boost;
} SdbQueryPlan;

/* Execute BM25 top-k over segments resident on this context's device.
 *
 * Mirrors ExecuteTopKWithCount (doc_collector.hpp:44-86): a two-phase
 * stats-then-collect execution (PreparePhase analogue,
 * duckdb_search_full_scan.cpp:1359-1384): BM25Stats idf/norm_const/
 * norm_length are computed on the host in double then narrowed to f32
 * exactly as bm25.cpp:288-306 does, then every segment is scored on the GPU
 * and a global top-k is selected.
 *
 * Results: hits[0..*out_count) sorted by (score desc, segment asc, doc asc);
 * *out_count = min(k, accepted); *total_matches = number of matching docs
 * (TotalMatches analogue, index/iterators.hpp:133). Acceptance mirrors the
 * collector: score > FLT_MIN (doc_collector.hpp:58 initial threshold).
 * Determinism: scores are bit-exact reproductions of the reference fp32
 * arithmetic with term-major merge order; ties at the k-th score resolve by
 * (segment, doc) ascending — a deterministic refinement of the reference's
 * unspecified nth_element tie order (DESIGN.md "Determinism").
 *
 * hits must have room for k entries. */
int sdb_gpu_execute_topk(SdbGpuCtx* ctx, SdbGpuSegment* const* segs,
                         uint32_t nsegs, const SdbQueryPlan* plan, uint32_t k,
                         SdbScoreDoc* hits, uint32_t* out_count,
                         uint64_t* total_matches);

/* Pipelined batch: nq queries of the SAME plan, each fully re-executed
 * (nothing cached between queries); query q+1's kernels overlap query
 * q's candidate readback and exact host select — the production QPS
 * shape of RunTopKScan's worker loop (duckdb_search_full_scan.cpp:
 * 1925-2000). hits: nq*k entries; out_counts/totals: nq entries.
 * Results of every query are identical to sdb_gpu_execute_topk's. */
int sdb_gpu_execute_topk_batch(SdbGpuCtx* ctx, SdbGpuSegment* const* segs,
                               uint32_t nsegs, const SdbQueryPlan* plan,
                               uint32_t k, uint32_t nq, SdbScoreDoc* hits,
                               uint32_t* out_counts, uint64_t* totals);

/* pipelined-batch form of the hybrid entry (same per-query semantics as
 * sdb_gpu_execute_topk_hybrid): bucket_counts/bucket_sums hold
 * nq * nbuckets rows, query-major. Bucket state is double-buffered by
 * query parity so query k+1's kernels overlap query k's readback. */
int sdb_gpu_execute_topk_hybrid_batch(
  SdbGpuCtx* ctx, SdbGpuSegment* const* segs, uint32_t nsegs,
  const SdbQueryPlan* plan, uint32_t k, int64_t flo, int64_t fhi,
  uint32_t nbuckets, uint32_t nq, int64_t* bucket_counts,
  int64_t* bucket_sums, SdbScoreDoc* hits, uint32_t* out_counts,
  uint64_t* totals);

/* CountFast — exact match count without scoring (docs-only decode;
 * DecideScanMode Count/CountFast, duckdb_search_full_scan.cpp:972). */
int sdb_gpu_execute_count(SdbGpuCtx* ctx, SdbGpuSegment* const* segs,
                          uint32_t nsegs, const SdbQueryPlan* plan,
                          uint64_t* total_matches);

/* Streaming scan — the RunStreamingScan / HitBatcher analogue
 * (duckdb_search_full_scan.cpp:2370, index/hit_batcher.hpp:39-190): emit
 * every matching doc id ascending into docs_out (up to cap) and, when
 * col_out != NULL and a column is attached, the gathered i64 value per hit.
 * *total_matches = full match count. Single segment. */
int sdb_gpu_execute_match_docs(SdbGpuCtx* ctx, SdbGpuSegment* seg,
                               const SdbQueryPlan* plan, uint32_t* docs_out,
                               int64_t* col_out, uint64_t cap,
                               uint64_t* out_count, uint64_t* total_matches);

/* Raw postings-block decode of one term into caller buffers (docs+freqs,
 * df entries each). Parity/diagnostic entry (mirrors
 * FormatTraits128::ReadBlockDelta/ReadBlock, format_block_128.hpp:446-636);
 * runs on the GPU. */
int sdb_gpu_decode_term(SdbGpuCtx* ctx, SdbGpuSegment* seg, uint32_t term_idx,
                        uint32_t* docs, uint32_t* freqs);

/* ---- hybrid: top-k AND pushed column predicates ----
 * MaybeWrapColFilter semantics (duckdb_search_full_scan.cpp TableFilter
 * wrap; index/table_filter_iterator.hpp:104-312 ColFilterChain): matches
 * survive only if EVERY attached-column predicate passes; the analytics
 * consumer's per-bucket COUNT/SUM aggregates over the survivors.
 * Columns attach to numbered slots (up to 4); the original
 * sdb_gpu_segment_attach_column is slot 0. */
#define SDB_MAX_FILTER_COLS 4

typedef enum SdbPredOp {
  SDB_PRED_NONE = 0,
  SDB_PRED_LT = 1,  /* col < v */
  SDB_PRED_GE = 2,  /* col >= v */
  SDB_PRED_BETWEEN = 3, /* lo <= col <= hi */
  SDB_PRED_EQ = 4,  /* col == v (table_filter_iterator.hpp typed compares) */
  /* bare null checks, evaluated on the validity plane alone
   * (table_filter_iterator.hpp:65-67 NullCheckKind) */
  SDB_PRED_ISNULL = 5,
  SDB_PRED_NOTNULL = 6,
  /* raw variable-width string columns (non-dictionary; SURVEY.md 8f row
   * 3, second half — the dictionary path shipped round 1):
   * SDB_PRED_STRMASK consumes the row bitmask a prior
   * sdb_gpu_strpred_mask() call computed for string slot `col`
   * (lo/hi/flo/fhi ignored); SDB_PRED_PREFIX is valid only as the `op`
   * of sdb_gpu_strpred_mask (starts-with). */
  SDB_PRED_STRMASK = 7,
  SDB_PRED_PREFIX = 8,
} SdbPredOp;

int sdb_gpu_segment_attach_column(SdbGpuCtx* ctx, SdbGpuSegment* seg,
                                  const int64_t* data);
int sdb_gpu_segment_attach_column_slot(SdbGpuCtx* ctx, SdbGpuSegment* seg,
                                       uint32_t slot, const int64_t* data);
/* per-doc f32 filter-boost column (doc_count+1 entries, 1-based docs);
 * consumed when SdbQueryPlan.filter_boost is set */
int sdb_gpu_segment_attach_boost(SdbGpuCtx* ctx, SdbGpuSegment* seg,
                                 const float* boost);
/* live-document bitmap — the deleted-docs mask the reference wraps around
 * every scan when a segment has deletes (seg.mask(it),
 * duckdb_search_full_scan.cpp:1898,2002,2226; Masked count mode :2475).
 * bit d of mask[d>>6] set = doc d live (docs 1..doc_count; bit 0 of word
 * 0 unused); (doc_count+64)/64 words. Once attached, every execute on the
 * segment sees only live docs (hits, counts, WAND thresholds, streaming
 * emission, hybrid aggregates). NULL detaches. */
int sdb_gpu_segment_attach_livemask(SdbGpuCtx* ctx, SdbGpuSegment* seg,
                                    const uint64_t* mask);

typedef struct SdbHybridPred {
  uint32_t slot;  /* attached column slot */
  SdbPredOp op;   /* LT / GE / BETWEEN (table_filter_iterator.hpp ops) */
  int64_t lo, hi; /* BETWEEN inclusive; LT/GE use lo */
} SdbHybridPred;

/* Single-predicate hybrid (BASELINE configs[3]): col0 BETWEEN [flo,fhi],
 * buckets over its value span. */
int sdb_gpu_execute_topk_hybrid(SdbGpuCtx* ctx, SdbGpuSegment* const* segs,
                                uint32_t nsegs, const SdbQueryPlan* plan,
                                uint32_t k, int64_t flo, int64_t fhi,
                                uint32_t nbuckets, int64_t* bucket_count,
                                int64_t* bucket_sum, SdbScoreDoc* hits,
                                uint32_t* out_count, uint64_t* total_matches);

/* Predicate-chain hybrid: preds[0] must be BETWEEN (it defines the bucket
 * span); preds[1..] are additional AND-ed predicates on any attached slot.
 * bucket_count/bucket_sum aggregate preds[0]'s column over full survivors. */
int sdb_gpu_execute_topk_hybrid_chain(
  SdbGpuCtx* ctx, SdbGpuSegment* const* segs, uint32_t nsegs,
  const SdbQueryPlan* plan, uint32_t k, const SdbHybridPred* preds,
  uint32_t npreds, uint32_t nbuckets, int64_t* bucket_count,
  int64_t* bucket_sum, SdbScoreDoc* hits, uint32_t* out_count,
  uint64_t* total_matches);

/* ---- columnar scan -> filter -> hash aggregate ---- */

/* Column layout: dense device-resident columns, or FoR/bitpack row groups
 * with min/max zonemaps (this repo's own codec — the reference's column
 * codecs live in the un-vendored DuckDB fork). */
typedef enum SdbColType {
  SDB_COL_I64 = 0,     /* dense i64 array */
  SDB_COL_F32 = 1,     /* dense f32 array */
  SDB_COL_I64_FOR = 2, /* FoR/bitpack row groups + zonemaps: the blob from
                          sdb_host_encode_col_i64 (data = blob pointer) */
} SdbColType;

typedef struct SdbColumnView {
  const void* data; /* HOST pointer at load time */
  uint64_t rows;
  SdbColType type;
} SdbColumnView;

typedef struct SdbGpuTable SdbGpuTable;

int sdb_gpu_table_load(SdbGpuCtx* ctx, const SdbColumnView* cols,
                       uint32_t ncols, uint64_t rows, SdbGpuTable** out);
int sdb_gpu_table_free(SdbGpuCtx* ctx, SdbGpuTable* tab);

/* Column validity (null) bitmap — SQL three-valued logic as the
 * reference's pushed filters implement it (null_semantics_fuzz.py;
 * table_filter_iterator.hpp NullCheckKind): a comparison predicate drops
 * NULL rows; SDB_PRED_ISNULL / SDB_PRED_NOTNULL evaluate the validity
 * plane alone; SUM aggregates skip NULL values while COUNT(*) counts the
 * row. bit r of bits[r>>6] set = row r VALID; (rows+63)/64 words. NULL
 * detaches (all valid). Validity on the GROUP-KEY column is rejected at
 * scan time (NULL-group semantics not implemented). */
int sdb_gpu_table_attach_validity(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                  uint32_t col, const uint64_t* bits);

/* Raw variable-width string column (non-dictionary; VERDICT r1 missing
 * #5): rows strings stored as offsets[rows+1] byte offsets into blob
 * (offsets monotone, offsets[rows] == blob_len; bytes arbitrary incl.
 * NUL — comparisons are lexicographic on unsigned bytes, i.e. memcmp
 * order, which for UTF-8 equals code-point order). Up to 4 slots per
 * table, independent of the i64/f32 column list. */
int sdb_gpu_table_attach_strcol(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                uint32_t slot, const uint64_t* offsets,
                                const uint8_t* blob, uint64_t blob_len);

/* Evaluate a string predicate over slot `slot` into a device-resident
 * row bitmask (bit r set = row matches). op: LT / GE / BETWEEN / EQ
 * (lexicographic vs lo[/hi]) or SDB_PRED_PREFIX (starts-with lo).
 * Literals are byte strings up to 63 bytes. The mask is then consumed by
 * scans via a predicate {col: slot, op: SDB_PRED_STRMASK}. Recompute the
 * mask before reuse with a different predicate (one mask per slot). */
int sdb_gpu_strpred_mask(SdbGpuCtx* ctx, SdbGpuTable* tab, uint32_t slot,
                         SdbPredOp op, const uint8_t* lo, uint32_t lo_len,
                         const uint8_t* hi, uint32_t hi_len);

/* FSST-style compressed string slot (SURVEY.md 8f row 3 "dict/FSST"):
 * rows hold code streams — code c < nsym expands to symbol bytes
 * symbols[sym_offsets[c]..sym_offsets[c+1]) (1..8 bytes each, nsym <=
 * 254, total symbol bytes <= 2048); code 255 escapes the next literal
 * byte. offsets[rows+1] index the ENCODED blob. Predicates on such a
 * slot (same sdb_gpu_strpred_mask call) decode on the fly against the
 * plain-byte literal — semantics identical to the uncompressed slot. */
int sdb_gpu_table_attach_strcol_fsst(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                     uint32_t slot,
                                     const uint64_t* offsets,
                                     const uint8_t* enc_blob,
                                     uint64_t enc_len,
                                     const uint8_t* symbols,
                                     const uint32_t* sym_offsets,
                                     uint32_t nsym);

typedef struct SdbPredSpec {
  uint32_t col;
  SdbPredOp op;
  int64_t ilo, ihi;
  float flo, fhi;
} SdbPredSpec;

typedef enum SdbAggOp {
  SDB_AGG_COUNT = 0,
  SDB_AGG_SUM_I64 = 1, /* exact wrap-around i64 sum */
  SDB_AGG_SUM_F64 = 2, /* f32 column summed in f64 */
} SdbAggOp;

typedef struct SdbAggSpec {
  uint32_t col; /* ignored for COUNT */
  SdbAggOp op;
} SdbAggSpec;

/* group_key column must be i64 with values in [0, ngroups).
 * out layout: for each group g, naggs consecutive values; COUNT/SUM_I64
 * results are int64 stored in the i64 field, SUM_F64 in the f64 field. */
typedef struct SdbAggResult {
  int64_t i64;
  double f64;
} SdbAggResult;

int sdb_gpu_scan_agg(SdbGpuCtx* ctx, SdbGpuTable* tab, uint32_t group_col,
                     uint32_t ngroups, const SdbPredSpec* preds,
                     uint32_t npreds, const SdbAggSpec* aggs, uint32_t naggs,
                     SdbAggResult* out /* ngroups*naggs */,
                     uint64_t* rows_passed);

/* General hash aggregate: ARBITRARY i64 group keys (sparse, any range) —
 * the north_star "hash-group-by with LDS-staged open-addressed buckets",
 * replacing the DuckDB PhysicalHashAggregate consumer
 * (duckdb_search_full_scan.cpp:2141-2171) for keys the dense perfect-hash
 * kernel (sdb_gpu_scan_agg) cannot take. Per-workgroup open-addressed LDS
 * table flushed once into a global open-addressed table; same predicate
 * pushdown and zonemap skips.
 *
 * max_groups bounds DISTINCT keys (<= 4M); more distinct keys (or a
 * pathological probe chain) returns SDB_ERR_OOM. Results: rows 0..
 * *ngroups_out sorted by key ascending; keys_out/out must hold max_groups
 * rows. Allocates/frees its device workspace per call (the dense kernel
 * remains the allocation-free hot path). */
int sdb_gpu_scan_agg_hash(SdbGpuCtx* ctx, SdbGpuTable* tab,
                          uint32_t group_col, uint64_t max_groups,
                          const SdbPredSpec* preds, uint32_t npreds,
                          const SdbAggSpec* aggs, uint32_t naggs,
                          int64_t* keys_out /* max_groups */,
                          SdbAggResult* out /* max_groups*naggs */,
                          uint64_t* ngroups_out, uint64_t* rows_passed);

/* GROUP BY string keys over an attached raw or FSST string slot (the
 * reference hands text group-by to its DuckDB fork's hash aggregate —
 * result-level parity). Each row's decoded bytes hash with FNV-1a 64 on
 * device; the hash feeds the same hash-aggregate core; keys_out holds
 * the group hashes sorted ascending as signed i64 (the core's
 * deterministic result order; hash -1 routes via the core's neg_acc
 * path like any other -1 key). The caller resolves
 * hashes back to strings and must verify injectivity over the column's
 * distinct strings (the python wrapper does both; collisions are
 * detected, never silent). */
int sdb_gpu_scan_agg_hash_str(SdbGpuCtx* ctx, SdbGpuTable* tab,
                              uint32_t str_slot, uint64_t max_groups,
                              const SdbPredSpec* preds, uint32_t npreds,
                              const SdbAggSpec* aggs, uint32_t naggs,
                              int64_t* keys_out, SdbAggResult* out,
                              uint64_t* ngroups_out,
                              uint64_t* rows_passed);

#ifdef __cplusplus
}
#endif
#endif /* SDB_GPU_H */
