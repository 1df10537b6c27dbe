/* sdb_oracle.c — CPU ORACLE for the SereneDB-AMD hot path.
 *
 * ============================ TEST INFRASTRUCTURE ============================
 * This library is TEST INFRASTRUCTURE ONLY. Per the project rules, only
 * tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may load or
 * call it — and only as the CHECKER (parity reference) or as the reported CPU
 * baseline. It is never the shipped product path; libsdb_gpu fails loudly if
 * the HIP extension is missing rather than falling back here.
 * ============================================================================
 *
 * This is a faithful scalar restatement of the reference algorithms, cited
 * per function below (paths relative to the reference tree):
 *  - block codec decode/encode: formats/posting/format_block_128.hpp
 *    :51-379 (write), :446-636 (read); delta-bitpack layout per vendored
 *    third_party/simdcomp (simdpackwithoutmaskd1/simdunpackd1,
 *    simdpackwithoutmask/simdunpack) — validated bit-for-bit against the
 *    reference's OWN simdcomp compiled into oracle/_ref (see Makefile);
 *    streamvbyte per the public Lemire 1234 format (submodule empty in the
 *    reference; version unpinned — see include/sdb_format.h).
 *  - postings iteration: formats/posting/iterator_doc.hpp:36-344.
 *  - BM25 stats + score kernel: search/bm25.cpp:279-306 (collect, double
 *    log1p -> f32), :89-109 (Bm25 score: c1 = norm_const +
 *    norm_length*norm; r = c0 - c0*c1/(c1+freq)), bm25.hpp:49-56 (stats).
 *  - windowed disjunction: search/block_disjunction.hpp:122-732 (4096-doc
 *    window: u64 mask[64] + f32 score window, children fill term-major);
 *    min-match via per-doc match counts; conjunction = min_match == nterms
 *    (search/conjunction.hpp:248-529 semantics).
 *  - top-k collector mechanics: NthPartitionScoreCollector
 *    (index/iterators.hpp:103-253): 2k-slot buffer, accept score >
 *    threshold, nth_element on overflow sets threshold = kth score.
 *  - driver: irs::ExecuteTopKWithCount (search/doc_collector.hpp:44-86):
 *    initial threshold FLT_MIN, final sort by score desc.
 *  - multithreaded baseline: RunTopKScan work claiming + shared kth-score
 *    CAS-max (server/connector/duckdb_search_full_scan.cpp:1868-1943).
 *
 * Determinism note (DESIGN.md "Determinism"): o_execute_topk (the parity
 * oracle) returns the EXACT top-k under the total order (score desc,
 * segment asc, doc asc) — a deterministic refinement of the reference's
 * nth_element-unspecified tie order. o_topk_collector_run exposes the raw
 * 2k/nth_element mechanics for validating transcribed reference fixtures.
 *
 * fp32 determinism: compiled with -ffp-contract=off; score expression is
 * plain IEEE fp32 mul/div/add/sub evaluated term-major, bit-identical to the
 * GPU kernel's (tests assert bitwise equality).
 */

#define _GNU_SOURCE
#include <math.h>
#include <pthread.h>
#include <stdatomic.h>
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#include "../include/sdb_format.h"

#define O_WINDOW 4096u /* block_disjunction.hpp:656 window: 4096 docs */

/* ------------------------------------------------------------------ */
/* codec — independent restatement                                     */
/* ------------------------------------------------------------------ */

static uint32_t o_bw(uint32_t v) { return v ? 32u - (uint32_t)__builtin_clz(v) : 0u; }

static uint32_t o_rd_word(const uint8_t* p, uint32_t widx) {
  uint32_t x;
  memcpy(&x, p + 4u * widx, 4);
  return x;
}

/* vertical unpack: value i -> lane i&3, group i>>2 (simdcomp layout) */
static void o_unpack128(const uint8_t* p, uint32_t bits, uint32_t* out) {
  const uint32_t mask = bits >= 32 ? 0xFFFFFFFFu : (1u << bits) - 1u;
  for (uint32_t i = 0; i < 128; ++i) {
    const uint32_t lane = i & 3u, grp = i >> 2;
    const uint32_t bit = grp * bits;
    const uint32_t w = bit >> 5, sh = bit & 31u;
    uint64_t v = (uint64_t)o_rd_word(p, w * 4 + lane) >> sh;
    if (sh + bits > 32)
      v |= (uint64_t)o_rd_word(p, (w + 1) * 4 + lane) << (32 - sh);
    out[i] = (uint32_t)v & mask;
  }
}

static void o_pack128(const uint32_t* in, uint32_t bits, uint8_t* p) {
  memset(p, 0, 16u * bits);
  for (uint32_t i = 0; i < 128; ++i) {
    const uint32_t lane = i & 3u, grp = i >> 2;
    const uint32_t bit = grp * bits;
    const uint32_t w = bit >> 5, sh = bit & 31u;
    uint64_t v = (uint64_t)in[i] << sh;
    uint32_t lo, hi;
    memcpy(&lo, p + 4u * (w * 4 + lane), 4);
    lo |= (uint32_t)v;
    memcpy(p + 4u * (w * 4 + lane), &lo, 4);
    if (sh + bits > 32) {
      memcpy(&hi, p + 4u * ((w + 1) * 4 + lane), 4);
      hi |= (uint32_t)(v >> 32);
      memcpy(p + 4u * ((w + 1) * 4 + lane), &hi, 4);
    }
  }
}

/* streamvbyte 1234 decode (public format): control bytes then data */
static uint32_t o_svb_decode(const uint8_t* in, uint32_t* out, uint32_t len) {
  const uint8_t* ctrl = in;
  const uint8_t* data = in + (len + 3) / 4;
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t code = (ctrl[i >> 2] >> ((i & 3) * 2)) & 3u;
    uint32_t v = 0;
    for (uint32_t b = 0; b <= code; ++b) v |= (uint32_t)(*data++) << (8 * b);
    out[i] = v;
  }
  return (uint32_t)(data - in);
}

/* ReadTailDelta (format_block_128.hpp:476-559) */
uint32_t o_decode_doc_block(const uint8_t* in, uint32_t len, uint32_t prev,
                            uint32_t* out) {
  const uint8_t* p = in;
  const uint32_t tag = *p++;
  if (tag == SDB_DE_VALUES) {
    memcpy(out, p, (size_t)len * 4);
    return 1 + len * 4;
  }
  if (tag >= SDB_DE_DELTA_ALL_SAME_08 && tag <= SDB_DE_DELTA_ALL_SAME_32) {
    uint32_t v = 0;
    uint32_t n = tag == SDB_DE_DELTA_ALL_SAME_08
                   ? 1u
                   : (tag == SDB_DE_DELTA_ALL_SAME_16 ? 2u : 4u);
    memcpy(&v, p, n);
    /* FillSameDelta (format_block_128.hpp:953-959): out[i]=prev+v+v*i */
    for (uint32_t i = 0; i < len; ++i) out[i] = prev + v + v * i;
    return 1 + n;
  }
  if (tag == SDB_DE_FOR_BITSET) {
    const uint32_t words = *p++;
    uint32_t n = 0;
    for (uint32_t w = 0; w < words; ++w) {
      uint64_t word;
      memcpy(&word, p + 8u * w, 8);
      while (word) {
        out[n++] = prev + w * 64u + (uint32_t)__builtin_ctzll(word);
        word &= word - 1;
      }
    }
    return 2 + words * 8;
  }
  if (tag == SDB_DE_STREAMVBYTE1234 || tag == SDB_DE_DELTA_STREAMVBYTE1234) {
    uint16_t size;
    memcpy(&size, p, 2);
    p += 2;
    o_svb_decode(p, out, len);
    if (tag == SDB_DE_DELTA_STREAMVBYTE1234) {
      uint32_t acc = prev;
      for (uint32_t i = 0; i < len; ++i) {
        acc += out[i];
        out[i] = acc;
      }
    }
    return 3 + size;
  }
  /* delta bitpack 2..31 — full blocks only (simdunpackd1) */
  {
    const uint32_t bits = tag - SDB_DE_DELTA_BITPACK_02 + 2;
    uint32_t tmp[128];
    o_unpack128(p, bits, tmp);
    uint32_t acc = prev;
    for (uint32_t i = 0; i < 128; ++i) {
      acc += tmp[i];
      out[i] = acc;
    }
    return 1 + 16 * bits;
  }
}

/* ReadTail (format_block_128.hpp:568-636) */
uint32_t o_decode_freq_block(const uint8_t* in, uint32_t len, uint32_t* out) {
  const uint8_t* p = in;
  const uint32_t tag = *p++;
  if (tag == SDB_E_VALUES) {
    memcpy(out, p, (size_t)len * 4);
    return 1 + len * 4;
  }
  if (tag >= SDB_E_ALL_SAME_08 && tag <= SDB_E_ALL_SAME_32) {
    uint32_t v = 0;
    uint32_t n =
      tag == SDB_E_ALL_SAME_08 ? 1u : (tag == SDB_E_ALL_SAME_16 ? 2u : 4u);
    memcpy(&v, p, n);
    for (uint32_t i = 0; i < len; ++i) out[i] = v;
    return 1 + n;
  }
  if (tag == SDB_E_STREAMVBYTE1234) {
    uint16_t size;
    memcpy(&size, p, 2);
    p += 2;
    o_svb_decode(p, out, len);
    return 3 + size;
  }
  {
    const uint32_t bits = tag - SDB_E_BITPACK_01 + 1;
    o_unpack128(p, bits, out);
    return 1 + 16 * bits;
  }
}

/* encode restatements (WriteTailDelta :57-242 / WriteTail :249-379) — used
 * only to cross-check the product encoder's bytes in tests. */
static uint32_t o_bs1234(uint32_t v) {
  return v < 256u ? 1u : v < 65536u ? 2u : v < (1u << 24) ? 3u : 4u;
}
static uint32_t o_svb_encode(const uint32_t* vals, uint32_t len, uint8_t* out) {
  const uint32_t groups = (len + 3) / 4;
  uint8_t* data = out + groups;
  memset(out, 0, groups);
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t n = o_bs1234(vals[i]);
    out[i >> 2] |= (uint8_t)((n - 1) << ((i & 3) * 2));
    uint32_t v = vals[i];
    for (uint32_t b = 0; b < n; ++b) { *data++ = (uint8_t)v; v >>= 8; }
  }
  return (uint32_t)(data - out);
}

uint32_t o_encode_doc_block(const uint32_t* in, uint32_t len, uint32_t prev,
                            uint8_t* out) {
  uint8_t best = SDB_DE_VALUES;
  uint32_t best_size = len * 4;
  int all_same = 1;
  const uint32_t for_max = in[len - 1] - prev;
  uint32_t dprev = prev, dmax = in[0] - prev;
  uint32_t svb = 2 + (len + 3) / 4, dsvb = svb;
  uint32_t deltas[128];
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t d = in[i] - dprev;
    dprev = in[i];
    deltas[i] = d;
    all_same &= (dmax == d);
    if (d > dmax) dmax = d;
    svb += o_bs1234(in[i]);
    dsvb += o_bs1234(d);
  }
  if (all_same) {
    best = dmax < 256 ? SDB_DE_DELTA_ALL_SAME_08
           : dmax < 65536 ? SDB_DE_DELTA_ALL_SAME_16
                          : SDB_DE_DELTA_ALL_SAME_32;
    best_size = dmax < 256 ? 1 : dmax < 65536 ? 2 : 4;
  } else {
    if (len == 128) {
      const uint32_t bits = o_bw(dmax);
      const uint32_t size = 16 * bits;
      if (size < best_size && bits <= 31) {
        best = (uint8_t)(SDB_DE_DELTA_BITPACK_02 + bits - 2);
        best_size = size;
      }
    } else {
      if (svb < best_size) { best = SDB_DE_STREAMVBYTE1234; best_size = svb; }
      if (dsvb < best_size) {
        best = SDB_DE_DELTA_STREAMVBYTE1234;
        best_size = dsvb;
      }
    }
    const uint32_t words = (for_max + 1 + 63) / 64;
    if (1 + words * 8 - 2 < best_size) {
      best = SDB_DE_FOR_BITSET;
      best_size = 1 + words * 8;
    }
  }
  uint8_t* p = out;
  *p++ = best;
  if (best == SDB_DE_VALUES) {
    memcpy(p, in, (size_t)len * 4);
    p += (size_t)len * 4;
  } else if (best <= SDB_DE_DELTA_ALL_SAME_32) {
    memcpy(p, &dmax, best_size);
    p += best_size;
  } else if (best == SDB_DE_FOR_BITSET) {
    const uint32_t words = (best_size - 1) / 8;
    uint64_t bs[64];
    memset(bs, 0, 8u * words);
    for (uint32_t i = 0; i < len; ++i) {
      const uint32_t v = in[i] - prev;
      bs[v >> 6] |= 1ull << (v & 63);
    }
    *p++ = (uint8_t)words;
    memcpy(p, bs, 8u * words);
    p += 8u * words;
  } else if (best == SDB_DE_STREAMVBYTE1234 ||
             best == SDB_DE_DELTA_STREAMVBYTE1234) {
    uint8_t buf[128 * 5 + 8];
    const uint32_t size = o_svb_encode(
      best == SDB_DE_STREAMVBYTE1234 ? in : deltas, len, buf);
    const uint16_t s16 = (uint16_t)size;
    memcpy(p, &s16, 2);
    p += 2;
    memcpy(p, buf, size);
    p += size;
  } else {
    o_pack128(deltas, best - SDB_DE_DELTA_BITPACK_02 + 2, p);
    p += best_size;
  }
  return (uint32_t)(p - out);
}

uint32_t o_encode_freq_block(const uint32_t* in, uint32_t len, uint8_t* out) {
  uint8_t best = SDB_E_VALUES;
  uint32_t best_size = len * 4;
  int all_same = 1;
  uint32_t max = in[0];
  uint32_t svb = 2 + (len + 3) / 4;
  for (uint32_t i = 0; i < len; ++i) {
    all_same &= (max == in[i]);
    if (in[i] > max) max = in[i];
    svb += o_bs1234(in[i]);
  }
  if (all_same) {
    best = max < 256 ? SDB_E_ALL_SAME_08
           : max < 65536 ? SDB_E_ALL_SAME_16
                         : SDB_E_ALL_SAME_32;
    best_size = max < 256 ? 1 : max < 65536 ? 2 : 4;
  } else if (len == 128) {
    const uint32_t bits = o_bw(max);
    const uint32_t size = 16 * bits;
    if (size < best_size && bits <= 31) {
      best = (uint8_t)(SDB_E_BITPACK_01 + bits - 1);
      best_size = size;
    }
  } else if (svb < best_size) {
    best = SDB_E_STREAMVBYTE1234;
    best_size = svb;
  }
  uint8_t* p = out;
  *p++ = best;
  if (best == SDB_E_VALUES) {
    memcpy(p, in, (size_t)len * 4);
    p += (size_t)len * 4;
  } else if (best <= SDB_E_ALL_SAME_32) {
    memcpy(p, &max, best_size);
    p += best_size;
  } else if (best == SDB_E_STREAMVBYTE1234) {
    uint8_t buf[128 * 5 + 8];
    const uint32_t size = o_svb_encode(in, len, buf);
    const uint16_t s16 = (uint16_t)size;
    memcpy(p, &s16, 2);
    p += 2;
    memcpy(p, buf, size);
    p += size;
  } else {
    o_pack128(in, best - SDB_E_BITPACK_01 + 1, p);
    p += best_size;
  }
  return (uint32_t)(p - out);
}

/* ------------------------------------------------------------------ */
/* segment view                                                        */
/* ------------------------------------------------------------------ */

int o_segment_parse(const void* blob, uint64_t size, SdbSegmentView* out) {
  if (!blob || size < sizeof(SdbSegHeader)) return -5;
  const SdbSegHeader* hdr = (const SdbSegHeader*)blob;
  if (hdr->magic != SDB_SEG_MAGIC || hdr->version < 1 ||
      hdr->version > 3 || hdr->version == 2 ||
      hdr->blob_size > size)
    return -5;
  const uint8_t* base = (const uint8_t*)blob;
  out->hdr = hdr;
  out->terms = (const SdbTermEntry*)(base + hdr->off_terms);
  out->desc = (const SdbBlockDesc*)(base + hdr->off_desc);
  out->norms = (const uint32_t*)(base + hdr->off_norms);
  out->payload = base + hdr->off_payload;
  return 0;
}

/* full-term decode (parity entry, mirrors sdb_gpu_decode_term) */
int o_decode_term(const SdbSegmentView* v, uint32_t term_idx, uint32_t* docs,
                  uint32_t* freqs) {
  if (term_idx >= v->hdr->nterms) return -1;
  const SdbTermEntry* te = &v->terms[term_idx];
  const uint8_t* pl = v->payload + te->payload_begin;
  uint32_t n = 0;
  for (uint64_t b = te->desc_begin; b < te->desc_end; ++b) {
    const SdbBlockDesc* d = &v->desc[b];
    o_decode_doc_block(pl + d->doc_off, d->len, d->prev_doc, docs + n);
    o_decode_freq_block(pl + d->freq_off, d->len, freqs + n);
    n += d->len;
  }
  return (int)n == (int)te->df ? 0 : -5;
}

/* ------------------------------------------------------------------ */
/* BM25 (search/bm25.cpp:279-306, :89-109)                             */
/* ------------------------------------------------------------------ */

void o_bm25_stats(uint64_t docs_with_field, uint64_t docs_with_term,
                  uint64_t total_term_freq, float k, float b, float* idf,
                  float* norm_const, float* norm_length) {
  *idf = (float)log1p(((double)(docs_with_field - docs_with_term) + 0.5) /
                      ((double)docs_with_term + 0.5));
  const float kb = k * b;
  if (b == 0.0f) {
    *norm_const = k;
    *norm_length = 0.0f;
    return;
  }
  *norm_const = k - kb;
  if (total_term_freq && docs_with_field) {
    const float avg_dl = (float)total_term_freq / (float)docs_with_field;
    *norm_length = kb / avg_dl;
  } else {
    *norm_length = kb;
  }
}

/* per-(doc,term) score — BM25 (bm25.cpp:89-109, num = boost*(k+1)*idf) or
 * TFIDF (tfidf.cpp:60-76, num = boost*idf) */
static inline float o_score_one(uint32_t scorer, float num, float norm_const,
                                float norm_length, uint32_t freq,
                                uint32_t norm) {
  if (scorer == 1) return num * sqrtf((float)freq);
  if (scorer == 2) return num * sqrtf((float)freq) / sqrtf((float)norm);
  const float c1 = norm_const + norm_length * (float)norm;
  return num - num * c1 / (c1 + (float)freq);
}

/* ------------------------------------------------------------------ */
/* query execution                                                     */
/* ------------------------------------------------------------------ */

typedef struct {
  float score;
  uint32_t doc;
  uint32_t seg;
} OScoreDoc;

typedef struct {
  uint32_t term_idx;
  float boost;
} OTermRef;

typedef struct {
  const SdbBlockDesc* d;
  const SdbBlockDesc* dend;
  const uint8_t* pl;
  uint32_t buf_docs[128];
  uint32_t buf_freqs[128];
  uint32_t buf_len, buf_pos;
  float num, nc, nl; /* prepared scorer constants */
  uint32_t scorer;   /* 0=BM25 (bm25.cpp:89-109), 1=TFIDF, 2=TFIDF+norms
                        (tfidf.cpp:60-76; idf per :148-151) */
} OCursor;

static void o_cursor_refill(OCursor* c) {
  if (c->d == c->dend) {
    c->buf_len = 0;
    c->buf_pos = 0;
    return;
  }
  o_decode_doc_block(c->pl + c->d->doc_off, c->d->len, c->d->prev_doc,
                     c->buf_docs);
  o_decode_freq_block(c->pl + c->d->freq_off, c->d->len, c->buf_freqs);
  c->buf_len = c->d->len;
  c->buf_pos = 0;
  c->d++;
}

/* ------------------------------------------------------------------ */
/* top-k collector — NthPartitionScoreCollector (iterators.hpp:103-253) */
/* ------------------------------------------------------------------ */

typedef struct {
  OScoreDoc* hits; /* 2k slots */
  uint32_t k;
  uint32_t it;      /* next write slot (_hits_it) */
  float* threshold; /* shared (&score_threshold) */
  uint64_t count;   /* TotalMatches (_count) */
  uint32_t seg;
} OCollector;

/* total order used everywhere ties must be deterministic:
 * (score desc, seg asc, doc asc) — DESIGN.md "Determinism" */
static int o_sd_cmp(const void* pa, const void* pb) {
  const OScoreDoc* a = (const OScoreDoc*)pa;
  const OScoreDoc* b = (const OScoreDoc*)pb;
  if (a->score != b->score) return a->score > b->score ? -1 : 1;
  if (a->seg != b->seg) return a->seg < b->seg ? -1 : 1;
  if (a->doc != b->doc) return a->doc < b->doc ? -1 : 1;
  return 0;
}

/* Push (iterators.hpp:216-229): on overflow order the 2k buffer descending
 * (qsort with the deterministic total order stands in for std::nth_element;
 * same retained top-k prefix, deterministic tie handling), reset write
 * cursor to k, threshold = score at index k (the reference's pivot). */
static void o_coll_push(OCollector* c, float score, uint32_t doc) {
  OScoreDoc* h = &c->hits[c->it];
  h->score = score;
  h->doc = doc;
  h->seg = c->seg;
  if (++c->it != 2 * c->k) return;
  qsort(c->hits, 2 * c->k, sizeof(OScoreDoc), o_sd_cmp);
  c->it = c->k;
  *c->threshold = c->hits[c->k].score;
}

static inline void o_coll_try(OCollector* c, float score, uint32_t doc) {
  if (score > *c->threshold) o_coll_push(c, score, doc);
}

/* ------------------------------------------------------------------ */
/* windowed execution over one doc range of one segment                */
/* (BlockDisjunction::RefillImpl block_disjunction.hpp:545-626 +       */
/*  AddWindow iterators.hpp:135-173)                                   */
/* ------------------------------------------------------------------ */

typedef struct {
  OScoreDoc* v;
  uint64_t n, cap;
} OCandVec;

static void o_cand_push(OCandVec* cv, float score, uint32_t doc,
                        uint32_t seg) {
  if (cv->n == cv->cap) {
    cv->cap = cv->cap ? cv->cap * 2 : 4096;
    cv->v = (OScoreDoc*)realloc(cv->v, cv->cap * sizeof(OScoreDoc));
  }
  cv->v[cv->n].score = score;
  cv->v[cv->n].doc = doc;
  cv->v[cv->n].seg = seg;
  cv->n++;
}

/* optional per-doc filter boost (HasFilterBoost scorer variants,
 * bm25.cpp:112-140 Bm1Boost / :69-109 Bm15/Bm25 boost[i]*num; modeled as
 * one shared per-doc multiplier applied per term contribution). NULL=off. */
static const float* o_fboost = NULL;
void o_set_filter_boost(const float* fb) { o_fboost = fb; }

/* optional live-document bitmap (deleted-docs mask, seg.mask(it) at
 * duckdb_search_full_scan.cpp:1898): bit d of word d>>6 = doc d live.
 * NULL = all live. */
static const uint64_t* o_live = NULL;
void o_set_live_mask(const uint64_t* m) { o_live = m; }

/* Process windows covering docs [range_lo, range_hi] (inclusive, 1-based,
 * local to the segment). Cursors must be positioned before range_lo.
 * Exactly one of coll / cands is non-NULL.
 * Returns the number of matching docs in the range. */
/* optional hybrid context (TableFilterDocIterator semantics,
 * index/table_filter_iterator.hpp:104-312: survivors of the pushed column
 * predicate; plus the analytics consumer's COUNT/SUM per bucket) */
typedef struct {
  const int64_t* col; /* indexed by doc (1-based), NULL = no filter */
  int64_t lo, hi;     /* BETWEEN, inclusive */
  uint32_t nbuckets;  /* bucket = (col-lo)*nbuckets/(hi-lo+1) */
  int64_t* bucket_count;
  int64_t* bucket_sum;
  /* extra AND-ed chain predicates (ColFilterChain,
   * table_filter_iterator.hpp:104-312); ops: 1=LT 2=GE 3=BETWEEN */
  uint32_t nextra;
  const int64_t* xcol[3];
  int xop[3];
  int64_t xlo[3], xhi[3];
} OHybrid;

static uint64_t o_exec_range(OCursor* cur, uint32_t nterms,
                             const uint32_t* norms, uint32_t min_match,
                             uint32_t range_lo, uint32_t range_hi,
                             OCollector* coll, OCandVec* cands, uint32_t seg,
                             const OHybrid* hy) {
  float score_win[O_WINDOW];
  uint8_t cnt_win[O_WINDOW];
  uint64_t mask[O_WINDOW / 64];
  uint64_t matches = 0;

  for (uint32_t lo = range_lo; lo <= range_hi && lo >= range_lo;
       lo += O_WINDOW) {
    const uint32_t hi =
      (range_hi - lo >= O_WINDOW - 1) ? lo + O_WINDOW - 1 : range_hi;
    const uint32_t wlen = hi - lo + 1;
    memset(score_win, 0, sizeof(float) * wlen);
    memset(cnt_win, 0, wlen);
    memset(mask, 0, sizeof(uint64_t) * ((wlen + 63) / 64));

    /* term-major fill: fp32 sum order is fixed (SURVEY.md §7 hard parts) */
    for (uint32_t t = 0; t < nterms; ++t) {
      OCursor* c = &cur[t];
      for (;;) {
        if (c->buf_pos == c->buf_len) {
          while (c->d != c->dend && c->d->last_doc < lo) c->d++; /* skip */
          if (c->d == c->dend) break;
          if (c->d->prev_doc >= hi) break; /* first doc > hi: not yet */
          o_cursor_refill(c);
        }
        uint32_t i = c->buf_pos;
        while (i < c->buf_len && c->buf_docs[i] < lo) ++i;
        while (i < c->buf_len && c->buf_docs[i] <= hi) {
          const uint32_t doc = c->buf_docs[i];
          const uint32_t off = doc - lo;
          /* filter boost folds into num BEFORE the score form, mirroring
           * the reference's op order (bm25.cpp: c0 = boost*num, then
           * c0 - c0*c1/(c1+freq); tfidf boost*num*sqrt(freq)) */
          const float nm = o_fboost ? c->num * o_fboost[doc] : c->num;
          float s1 = o_score_one(c->scorer, nm, c->nc, c->nl,
                                 c->buf_freqs[i], norms[doc]);
          score_win[off] += s1;
          const uint8_t cc = ++cnt_win[off];
          if (cc == min_match) mask[off >> 6] |= 1ull << (off & 63);
          ++i;
        }
        c->buf_pos = i;
        if (i == c->buf_len) continue; /* block drained: maybe next block */
        break;                         /* next doc beyond window */
      }
    }

    /* AddWindow (iterators.hpp:135-173): emit set bits */
    for (uint32_t w = 0; w < (wlen + 63) / 64; ++w) {
      uint64_t word = mask[w];
      if (!word) continue;
      matches += (uint64_t)__builtin_popcountll(word);
      const uint32_t base = lo + w * 64;  /* hy filter decrements below */
      while (word) {
        const uint32_t bit = (uint32_t)__builtin_ctzll(word);
        word &= word - 1;
        const uint32_t doc = base + bit;
        if (o_live && !((o_live[doc >> 6] >> (doc & 63u)) & 1ull)) {
          --matches; /* masked doc: invisible (popcount pre-counted it) */
          continue;
        }
        if (hy && hy->col) {
          const int64_t v = hy->col[doc];
          if (v < hy->lo || v > hy->hi) {
            --matches; /* popcount above pre-counted this doc */
            continue;
          }
          int pass = 1;
          for (uint32_t x = 0; x < hy->nextra; ++x) {
            const int64_t xv = hy->xcol[x][doc];
            if (hy->xop[x] == 1) pass &= xv < hy->xlo[x];
            else if (hy->xop[x] == 2) pass &= xv >= hy->xlo[x];
            else pass &= (xv >= hy->xlo[x]) && (xv <= hy->xhi[x]);
          }
          if (!pass) {
            --matches;
            continue;
          }
          const uint64_t span = (uint64_t)(hy->hi - hy->lo) + 1;
          uint32_t bkt =
            (uint32_t)(((uint64_t)(v - hy->lo) * hy->nbuckets) / span);
          if (bkt >= hy->nbuckets) bkt = hy->nbuckets - 1;
          hy->bucket_count[bkt] += 1;
          hy->bucket_sum[bkt] += v;
        }
        const float s = score_win[w * 64 + bit];
        if (coll) {
          coll->count++;
          o_coll_try(coll, s, doc);
        } else {
          o_cand_push(cands, s, doc, seg);
        }
      }
    }
  }
  return matches;
}

/* ------------------------------------------------------------------ */
/* drivers                                                             */
/* ------------------------------------------------------------------ */

typedef struct {
  const void* blob;
  uint64_t size;
} OSegBlob;

/* Prepare cursors + scorer constants for one segment.
 * Global stats (sharded execution) may be injected; zeros mean "derive from
 * the provided segments" (ExecuteTopKWithCount single-node semantics). */
/* binary search: first block with last_doc >= lo (the skip list's
 * SeekToBlock analogue, skip_list.hpp) — the multithreaded baseline seeks
 * each work chunk in O(log blocks); a linear walk from the term start made
 * per-chunk positioning quadratic and understated the CPU baseline ~50x. */
static const SdbBlockDesc* o_seek_block(const SdbBlockDesc* b,
                                        const SdbBlockDesc* e, uint32_t lo) {
  while (b < e) {
    const SdbBlockDesc* m = b + (e - b) / 2;
    if (m->last_doc < lo)
      b = m + 1;
    else
      e = m;
  }
  return b;
}

static uint32_t o_scorer_kind = 0; /* set per call by the drivers (the C
  API keeps single-threaded-per-call semantics; sdb_oracle is test infra) */


static int o_prep_cursors(const SdbSegmentView* v, const uint32_t* term_idx,
                          const float* boosts, uint32_t nterms, float k1,
                          float b, uint64_t g_dwf, const uint64_t* g_dwt,
                          uint64_t g_ttf, OCursor* cur) {
  for (uint32_t t = 0; t < nterms; ++t) {
    if (term_idx[t] >= v->hdr->nterms) return -1;
    const SdbTermEntry* te = &v->terms[term_idx[t]];
    OCursor* c = &cur[t];
    memset(c, 0, sizeof(*c));
    c->d = v->desc + te->desc_begin;
    c->dend = v->desc + te->desc_end;
    c->pl = v->payload + te->payload_begin;
    float idf, nc, nl;
    const uint64_t dwf = g_dwf ? g_dwf : v->hdr->docs_with_field;
    const uint64_t dwt = g_dwt ? g_dwt[t] : te->df;
    const uint64_t ttf = g_ttf ? g_ttf : v->hdr->total_term_freq;
    c->scorer = o_scorer_kind;
    if (dwt == 0) { /* term absent everywhere: no iterator */
      c->d = c->dend;
      c->num = 0.0f;
      c->nc = k1;
      c->nl = 0.0f;
      continue;
    }
    if (o_scorer_kind != 0) { /* TFIDF::collect (tfidf.cpp:148-151) */
      idf = (float)log1p(((double)dwf + 1.0) / ((double)dwt + 1.0));
      c->num = boosts[t] * idf;
      c->nc = 0.0f;
      c->nl = 0.0f;
      continue;
    }
    o_bm25_stats(dwf, dwt, ttf, k1, b, &idf, &nc, &nl);
    c->num = boosts[t] * (k1 + 1.0f) * idf; /* bm25.cpp Bm25Score::num */
    c->nc = nc;
    c->nl = nl;
    /* BM1 (k == 0, bm25.cpp:112-140 Bm1Score + :333-336 dispatch): without
     * a filter boost every score is 0 (memset), so nothing beats the
     * collector's FLT_MIN threshold; num = 0 reproduces that exactly.
     * WITH a filter boost: score = fb * num (Bm1Boost), freq-independent,
     * which the generic formula already yields at c1 == 0. */
    if (k1 == 0.0f && !o_fboost) c->num = 0.0f;
  }
  return 0;
}

/* scorer selection for subsequent o_execute_* calls (0=BM25, 1=TFIDF,
 * 2=TFIDF with norms) */
void o_set_scorer(uint32_t kind) { o_scorer_kind = kind; }

/* Cross-segment stats merge — the reference merges Field/TermCollector
 * counters over ALL segments before BM25::collect runs (PrepareCollector
 * Finish, search/collectors.cpp; PreparePhase barrier,
 * duckdb_search_full_scan.cpp:1359-1384): idf/avgDL are GLOBAL. When the
 * caller passes no explicit global stats, derive them by summing the
 * provided segments. Returns 0 and fills dwf/ttf/dwt[nterms]. */
static int o_global_stats(const OSegBlob* segs, uint32_t nsegs,
                          const uint32_t* term_idx, uint32_t nterms,
                          uint64_t* dwf, uint64_t* ttf, uint64_t* dwt) {
  *dwf = 0;
  *ttf = 0;
  memset(dwt, 0, sizeof(uint64_t) * nterms);
  for (uint32_t s = 0; s < nsegs; ++s) {
    SdbSegmentView v;
    int rc = o_segment_parse(segs[s].blob, segs[s].size, &v);
    if (rc) return rc;
    *dwf += v.hdr->docs_with_field;
    *ttf += v.hdr->total_term_freq;
    for (uint32_t t = 0; t < nterms; ++t) {
      if (term_idx[t] >= v.hdr->nterms) return -1;
      dwt[t] += v.terms[term_idx[t]].df;
    }
  }
  return 0;
}

/* EXACT top-k (the parity oracle): gathers every matching (doc,score), then
 * selects under the deterministic total order. Mirrors
 * ExecuteTopKWithCount's results up to nth_element tie refinement. */
int o_execute_topk(const OSegBlob* segs, uint32_t nsegs,
                   const uint32_t* term_idx, const float* boosts,
                   uint32_t nterms, uint32_t min_match, float k1, float b,
                   uint64_t g_dwf, const uint64_t* g_dwt, uint64_t g_ttf,
                   uint32_t k, OScoreDoc* hits, uint32_t* out_count,
                   uint64_t* total_matches) {
  OCandVec cands = {0, 0, 0};
  uint64_t matches = 0;
  uint64_t auto_dwf = 0, auto_ttf = 0;
  uint64_t* auto_dwt = (uint64_t*)malloc(sizeof(uint64_t) * nterms);
  if (g_dwf == 0) {
    int rc = o_global_stats(segs, nsegs, term_idx, nterms, &auto_dwf,
                            &auto_ttf, auto_dwt);
    if (rc) { free(auto_dwt); return rc; }
    g_dwf = auto_dwf;
    g_ttf = auto_ttf;
    g_dwt = auto_dwt;
  }
  OCursor* cur = (OCursor*)malloc(sizeof(OCursor) * nterms);
  for (uint32_t s = 0; s < nsegs; ++s) {
    SdbSegmentView v;
    int rc = o_segment_parse(segs[s].blob, segs[s].size, &v);
    if (rc) { free(cur); free(cands.v); free(auto_dwt); return rc; }
    rc = o_prep_cursors(&v, term_idx, boosts, nterms, k1, b, g_dwf, g_dwt,
                        g_ttf, cur);
    if (rc) { free(cur); free(cands.v); free(auto_dwt); return rc; }
    matches += o_exec_range(cur, nterms, v.norms, min_match ? min_match : 1,
                            1, v.hdr->doc_count, NULL, &cands, s, NULL);
  }
  free(cur);
  free(auto_dwt);
  /* threshold semantics: accept score > FLT_MIN (doc_collector.hpp:58) */
  uint64_t n = 0;
  const float kFltMin = 1.17549435e-38f;
  for (uint64_t i = 0; i < cands.n; ++i)
    if (cands.v[i].score > kFltMin) cands.v[n++] = cands.v[i];
  qsort(cands.v, n, sizeof(OScoreDoc), o_sd_cmp);
  const uint32_t kk = n < k ? (uint32_t)n : k;
  memcpy(hits, cands.v, sizeof(OScoreDoc) * kk);
  *out_count = kk;
  *total_matches = matches;
  free(cands.v);
  return 0;
}

/* HYBRID exact top-k: BM25 match AND col BETWEEN [lo,hi], plus per-bucket
 * COUNT/SUM over the surviving matches (BASELINE configs[3]; reference
 * semantics: MaybeWrapColFilter + TableFilterDocIterator::Collect,
 * duckdb_search_full_scan.cpp:1900-1912, table_filter_iterator.hpp:229-312,
 * aggregate consumer external). col indexed by LOCAL doc id of the single
 * provided segment. */
static int o_hybrid_impl(const void* blob, uint64_t size,
                         const uint32_t* term_idx, const float* boosts,
                         uint32_t nterms, uint32_t min_match, float k1,
                         float b, uint64_t g_dwf, const uint64_t* g_dwt,
                         uint64_t g_ttf, uint32_t k, const int64_t* col,
                         int64_t flo, int64_t fhi, uint32_t nbuckets,
                         const int64_t* const* chain_cols,
                         const int* chain_ops, const int64_t* chain_los,
                         const int64_t* chain_his, uint32_t ncols_chain,
                         int64_t* bucket_count, int64_t* bucket_sum,
                         OScoreDoc* hits, uint32_t* out_count,
                         uint64_t* total_matches) {
  SdbSegmentView v;
  int rc = o_segment_parse(blob, size, &v);
  if (rc) return rc;
  OSegBlob sb = {blob, size};
  uint64_t auto_dwf = 0, auto_ttf = 0;
  uint64_t* auto_dwt = (uint64_t*)malloc(sizeof(uint64_t) * nterms);
  if (g_dwf == 0) {
    rc = o_global_stats(&sb, 1, term_idx, nterms, &auto_dwf, &auto_ttf,
                        auto_dwt);
    if (rc) { free(auto_dwt); return rc; }
    g_dwf = auto_dwf;
    g_ttf = auto_ttf;
    g_dwt = auto_dwt;
  }
  OCursor* cur = (OCursor*)malloc(sizeof(OCursor) * nterms);
  rc = o_prep_cursors(&v, term_idx, boosts, nterms, k1, b, g_dwf, g_dwt,
                      g_ttf, cur);
  if (rc) { free(cur); free(auto_dwt); return rc; }
  memset(bucket_count, 0, sizeof(int64_t) * nbuckets);
  memset(bucket_sum, 0, sizeof(int64_t) * nbuckets);
  OHybrid hy;
  memset(&hy, 0, sizeof(hy));
  hy.col = col;
  hy.lo = flo;
  hy.hi = fhi;
  hy.nbuckets = nbuckets;
  hy.bucket_count = bucket_count;
  hy.bucket_sum = bucket_sum;
  for (uint32_t x = 0; x + 1 < ncols_chain; ++x) {
    hy.xcol[x] = chain_cols[x + 1];
    hy.xop[x] = chain_ops[x + 1];
    hy.xlo[x] = chain_los[x + 1];
    hy.xhi[x] = chain_his[x + 1];
  }
  hy.nextra = ncols_chain ? ncols_chain - 1 : 0;
  OCandVec cands = {0, 0, 0};
  uint64_t matches =
    o_exec_range(cur, nterms, v.norms, min_match ? min_match : 1, 1,
                 v.hdr->doc_count, NULL, &cands, 0, &hy);
  free(cur);
  free(auto_dwt);
  uint64_t n = 0;
  const float kFltMin = 1.17549435e-38f;
  for (uint64_t i = 0; i < cands.n; ++i)
    if (cands.v[i].score > kFltMin) cands.v[n++] = cands.v[i];
  qsort(cands.v, n, sizeof(OScoreDoc), o_sd_cmp);
  const uint32_t kk = n < k ? (uint32_t)n : k;
  memcpy(hits, cands.v, sizeof(OScoreDoc) * kk);
  *out_count = kk;
  *total_matches = matches;
  free(cands.v);
  return 0;
}

int o_execute_topk_hybrid(const void* blob, uint64_t size,
                          const uint32_t* term_idx, const float* boosts,
                          uint32_t nterms, uint32_t min_match, float k1,
                          float b, uint64_t g_dwf, const uint64_t* g_dwt,
                          uint64_t g_ttf, uint32_t k, const int64_t* col,
                          int64_t flo, int64_t fhi, uint32_t nbuckets,
                          int64_t* bucket_count, int64_t* bucket_sum,
                          OScoreDoc* hits, uint32_t* out_count,
                          uint64_t* total_matches) {
  return o_hybrid_impl(blob, size, term_idx, boosts, nterms, min_match, k1,
                       b, g_dwf, g_dwt, g_ttf, k, col, flo, fhi, nbuckets,
                       NULL, NULL, NULL, NULL, 0, bucket_count, bucket_sum,
                       hits, out_count, total_matches);
}

/* Chain form: cols[0] carries the BETWEEN (los[0]..his[0]) bucket span
 * (ops[0] must be 3); cols[1..ncols) AND-narrow with ops LT/GE/BETWEEN. */
int o_execute_topk_hybrid_chain(
  const void* blob, uint64_t size, const uint32_t* term_idx,
  const float* boosts, uint32_t nterms, uint32_t min_match, float k1,
  float b, uint32_t k, const int64_t* const* cols, const int* ops,
  const int64_t* los, const int64_t* his, uint32_t ncols, uint32_t nbuckets,
  int64_t* bucket_count, int64_t* bucket_sum, OScoreDoc* hits,
  uint32_t* out_count, uint64_t* total_matches) {
  if (ncols == 0 || ncols > 4 || ops[0] != 3) return -1;
  return o_hybrid_impl(blob, size, term_idx, boosts, nterms, min_match, k1,
                       b, 0, NULL, 0, k, cols[0], los[0], his[0], nbuckets,
                       cols, ops, los, his, ncols, bucket_count, bucket_sum,
                       hits, out_count, total_matches);
}

/* STREAMING scan: emit every matching doc id ascending, plus (optionally)
 * the gathered values of an attached i64 column — the RunStreamingScan /
 * HitBatcher analogue (duckdb_search_full_scan.cpp:2370,
 * index/hit_batcher.hpp:39-190: windows of hit doc-ids -> column gather ->
 * DataChunk). Single segment. Returns up to cap docs; *total_matches is the
 * full count (emission stops at cap but counting continues). */
int o_execute_match_docs(const void* blob, uint64_t size,
                         const uint32_t* term_idx, const float* boosts,
                         uint32_t nterms, uint32_t min_match, float k1,
                         float b, const int64_t* col, int64_t* col_out,
                         uint32_t* docs_out, uint64_t cap,
                         uint64_t* out_count, uint64_t* total_matches) {
  SdbSegmentView v;
  int rc = o_segment_parse(blob, size, &v);
  if (rc) return rc;
  OSegBlob sb = {blob, size};
  uint64_t dwf, ttf;
  uint64_t* dwt = (uint64_t*)malloc(sizeof(uint64_t) * nterms);
  rc = o_global_stats(&sb, 1, term_idx, nterms, &dwf, &ttf, dwt);
  if (rc) { free(dwt); return rc; }
  OCursor* cur = (OCursor*)malloc(sizeof(OCursor) * nterms);
  rc = o_prep_cursors(&v, term_idx, boosts, nterms, k1, b, dwf, dwt, ttf,
                      cur);
  if (rc) { free(cur); free(dwt); return rc; }
  OCandVec cands = {0, 0, 0};
  uint64_t matches =
    o_exec_range(cur, nterms, v.norms, min_match ? min_match : 1, 1,
                 v.hdr->doc_count, NULL, &cands, 0, NULL);
  /* cands are window-ordered = doc-ascending already (single segment) */
  uint64_t n = cands.n < cap ? cands.n : cap;
  for (uint64_t i = 0; i < n; ++i) {
    docs_out[i] = cands.v[i].doc;
    if (col && col_out) col_out[i] = col[cands.v[i].doc];
  }
  *out_count = n;
  *total_matches = matches;
  free(cands.v);
  free(cur);
  free(dwt);
  return 0;
}

/* MECHANICS emulation (2k buffer + threshold), single-threaded — validates
 * transcribed reference fixtures and is the timed single-thread CPU path. */
int o_execute_topk_mech(const OSegBlob* segs, uint32_t nsegs,
                        const uint32_t* term_idx, const float* boosts,
                        uint32_t nterms, uint32_t min_match, float k1, float b,
                        uint64_t g_dwf, const uint64_t* g_dwt, uint64_t g_ttf,
                        uint32_t k, OScoreDoc* hits2k /* 2k slots */,
                        uint32_t* out_count, uint64_t* total_matches) {
  uint64_t auto_dwf = 0, auto_ttf = 0;
  uint64_t* auto_dwt = (uint64_t*)malloc(sizeof(uint64_t) * nterms);
  if (g_dwf == 0) {
    int rc = o_global_stats(segs, nsegs, term_idx, nterms, &auto_dwf,
                            &auto_ttf, auto_dwt);
    if (rc) { free(auto_dwt); return rc; }
    g_dwf = auto_dwf;
    g_ttf = auto_ttf;
    g_dwt = auto_dwt;
  }
  float threshold = 1.17549435e-38f; /* FLT_MIN, doc_collector.hpp:58 */
  OCollector coll;
  coll.hits = hits2k;
  coll.k = k;
  coll.it = 0;
  coll.threshold = &threshold;
  coll.count = 0;
  OCursor* cur = (OCursor*)malloc(sizeof(OCursor) * nterms);
  for (uint32_t s = 0; s < nsegs; ++s) {
    SdbSegmentView v;
    int rc = o_segment_parse(segs[s].blob, segs[s].size, &v);
    if (rc) { free(cur); free(auto_dwt); return rc; }
    rc = o_prep_cursors(&v, term_idx, boosts, nterms, k1, b, g_dwf, g_dwt,
                        g_ttf, cur);
    if (rc) { free(cur); free(auto_dwt); return rc; }
    coll.seg = s;
    o_exec_range(cur, nterms, v.norms, min_match ? min_match : 1, 1,
                 v.hdr->doc_count, &coll, NULL, s, NULL);
  }
  free(cur);
  free(auto_dwt);
  qsort(hits2k, coll.it, sizeof(OScoreDoc), o_sd_cmp); /* :82-85 final sort */
  *out_count = coll.it;
  *total_matches = coll.count;
  return 0;
}

/* ------------------------------------------------------------------ */
/* multithreaded CPU baseline — RunTopKScan shape                      */
/* (duckdb_search_full_scan.cpp:1868-1943: atomic work claiming,       */
/*  per-thread 2k buffers, shared kth-score CAS max :1918-1921)        */
/* ------------------------------------------------------------------ */

typedef struct {
  const SdbSegmentView* v;
  const uint32_t* term_idx;
  const float* boosts;
  uint32_t nterms, min_match, k;
  float k1, b;
  uint64_t g_dwf, g_ttf;
  const uint64_t* g_dwt;
  uint32_t chunk_docs;
  _Atomic uint32_t* next_chunk;
  uint32_t nchunks;
  _Atomic uint32_t* g_thresh_bits; /* float bits; scores >= 0 so ordered */
  OScoreDoc* hits2k;               /* this thread's buffer */
  uint32_t out_count;
  uint64_t matches;
  /* bench amortization: run the whole query `iters` times inside the pool
   * (256 pthread_create/join per 0.4 ms query otherwise dominates) */
  uint32_t iters;
  uint32_t thread_id;
  pthread_barrier_t* barrier;
} OMtArg;

static float o_atomic_thresh_get(_Atomic uint32_t* p) {
  uint32_t b = atomic_load_explicit(p, memory_order_relaxed);
  float f;
  memcpy(&f, &b, 4);
  return f;
}
static void o_atomic_thresh_max(_Atomic uint32_t* p, float val) {
  uint32_t nb;
  memcpy(&nb, &val, 4);
  uint32_t ob = atomic_load_explicit(p, memory_order_relaxed);
  for (;;) {
    float of;
    memcpy(&of, &ob, 4);
    if (val <= of) return;
    if (atomic_compare_exchange_weak_explicit(p, &ob, nb,
                                              memory_order_relaxed,
                                              memory_order_relaxed))
      return;
  }
}

static void* o_mt_worker(void* argp) {
  OMtArg* a = (OMtArg*)argp;
  OCursor* cur = (OCursor*)malloc(sizeof(OCursor) * a->nterms);
  OCollector coll;
  uint64_t matches = 0;
  float threshold = 0.0f;
  for (uint32_t it = 0; it < a->iters; ++it) {
  if (a->barrier) pthread_barrier_wait(a->barrier);
  if (a->thread_id == 0) {
    atomic_store_explicit(a->next_chunk, 0u, memory_order_relaxed);
    float f0 = 1.17549435e-38f;
    uint32_t b0;
    memcpy(&b0, &f0, 4);
    atomic_store_explicit(a->g_thresh_bits, b0, memory_order_relaxed);
  }
  if (a->barrier) pthread_barrier_wait(a->barrier);
  threshold = o_atomic_thresh_get(a->g_thresh_bits);
  coll.hits = a->hits2k;
  coll.k = a->k;
  coll.it = 0;
  coll.threshold = &threshold;
  coll.count = 0;
  coll.seg = 0;
  matches = 0;
  for (;;) {
    const uint32_t chunk = atomic_fetch_add_explicit(a->next_chunk, 1,
                                                     memory_order_relaxed);
    if (chunk >= a->nchunks) break;
    const uint32_t lo = 1 + chunk * a->chunk_docs;
    uint32_t hi = lo + a->chunk_docs - 1;
    if (hi > a->v->hdr->doc_count) hi = a->v->hdr->doc_count;
    o_prep_cursors(a->v, a->term_idx, a->boosts, a->nterms, a->k1, a->b,
                   a->g_dwf, a->g_dwt, a->g_ttf, cur);
    for (uint32_t t = 0; t < a->nterms; ++t)
      cur[t].d = o_seek_block(cur[t].d, cur[t].dend, lo);
    /* pull the shared threshold before the chunk; publish after
     * (duckdb_search_full_scan.cpp:1918-1921 CAS-max mirror) */
    const float g = o_atomic_thresh_get(a->g_thresh_bits);
    if (g > threshold) threshold = g;
    matches += o_exec_range(cur, a->nterms, a->v->norms,
                            a->min_match ? a->min_match : 1, lo, hi, &coll,
                            NULL, 0, NULL);
    o_atomic_thresh_max(a->g_thresh_bits, threshold);
  }
  }  /* iters */
  free(cur);
  a->out_count = coll.it;
  a->matches = matches;
  return NULL;
}

/* Timed multithreaded baseline over ONE segment blob. Returns matches and
 * fills hits (k slots) with the final merged top-k. */
int o_execute_topk_mt_iters(const void* blob, uint64_t size,
                            const uint32_t* term_idx, const float* boosts,
                            uint32_t nterms, uint32_t min_match, float k1,
                            float b, uint64_t g_dwf, const uint64_t* g_dwt,
                            uint64_t g_ttf, uint32_t k, uint32_t nthreads,
                            uint32_t iters, OScoreDoc* hits,
                            uint32_t* out_count, uint64_t* total_matches) {
  SdbSegmentView v;
  int rc = o_segment_parse(blob, size, &v);
  if (rc) return rc;
  if (nthreads == 0) nthreads = 1;
  const uint32_t chunk_docs = 8 * O_WINDOW; /* 32768-doc work units */
  const uint32_t nchunks = (v.hdr->doc_count + chunk_docs - 1) / chunk_docs;
  _Atomic uint32_t next_chunk = 0;
  _Atomic uint32_t thresh_bits;
  {
    float f = 1.17549435e-38f;
    uint32_t bits;
    memcpy(&bits, &f, 4);
    atomic_store(&thresh_bits, bits);
  }
  OMtArg* args = (OMtArg*)calloc(nthreads, sizeof(OMtArg));
  pthread_t* th = (pthread_t*)malloc(sizeof(pthread_t) * nthreads);
  OScoreDoc* bufs = (OScoreDoc*)malloc(sizeof(OScoreDoc) * 2ull * k * nthreads);
  pthread_barrier_t barrier;
  pthread_barrier_init(&barrier, NULL, nthreads);
  if (iters == 0) iters = 1;
  for (uint32_t t = 0; t < nthreads; ++t) {
    args[t] = (OMtArg){&v,       term_idx, boosts,     nterms,
                       min_match, k,        k1,         b,
                       g_dwf,     g_ttf,    g_dwt,      chunk_docs,
                       &next_chunk, nchunks, &thresh_bits,
                       bufs + 2ull * k * t, 0, 0, iters, t, &barrier};
    pthread_create(&th[t], NULL, o_mt_worker, &args[t]);
  }
  uint64_t matches = 0;
  OCandVec all = {0, 0, 0};
  for (uint32_t t = 0; t < nthreads; ++t) {
    pthread_join(th[t], NULL);
    matches += args[t].matches;
    for (uint32_t i = 0; i < args[t].out_count; ++i)
      o_cand_push(&all, bufs[2ull * k * t + i].score,
                  bufs[2ull * k * t + i].doc, 0);
  }
  const float kFltMin = 1.17549435e-38f;
  uint64_t n = 0;
  for (uint64_t i = 0; i < all.n; ++i)
    if (all.v[i].score > kFltMin) all.v[n++] = all.v[i];
  qsort(all.v, n, sizeof(OScoreDoc), o_sd_cmp);
  const uint32_t kk = n < k ? (uint32_t)n : k;
  memcpy(hits, all.v, sizeof(OScoreDoc) * kk);
  *out_count = kk;
  *total_matches = matches;
  free(all.v);
  free(bufs);
  free(th);
  free(args);
  pthread_barrier_destroy(&barrier);
  return 0;
}

int o_execute_topk_mt(const void* blob, uint64_t size,
                      const uint32_t* term_idx, const float* boosts,
                      uint32_t nterms, uint32_t min_match, float k1, float b,
                      uint64_t g_dwf, const uint64_t* g_dwt, uint64_t g_ttf,
                      uint32_t k, uint32_t nthreads, OScoreDoc* hits,
                      uint32_t* out_count, uint64_t* total_matches) {
  return o_execute_topk_mt_iters(blob, size, term_idx, boosts, nterms,
                                 min_match, k1, b, g_dwf, g_dwt, g_ttf, k,
                                 nthreads, 1, hits, out_count,
                                 total_matches);
}

/* ------------------------------------------------------------------ */
/* columnar scan -> filter -> group-by reference (config 3)            */
/* restates FullScanner::Scan + ColFilterChain narrowing semantics     */
/* (server/connector/full_scanner.h:40-90,                             */
/*  index/table_filter_iterator.hpp:104-227) feeding a hash aggregate  */
/* (external DuckDB PhysicalHashAggregate; parity at result level,     */
/*  SURVEY.md §8c). COUNT/SUM(i64) exact; SUM(f32) accumulated in f64  */
/*  sequentially (deterministic oracle order).                         */
/* ------------------------------------------------------------------ */
/* multithreaded scan baseline: row chunks over a thread pool, per-thread
 * group arrays merged in thread-index order (deterministic i64; f64 merge
 * order fixed). iters repeats the scan inside the pool (thread-spawn
 * amortization, as in the top-k baseline). */
typedef struct {
  const int64_t* keys;
  const int64_t* v1;
  const float* v2;
  uint64_t rows;
  uint32_t ngroups;
  int pred_op;
  int64_t lo, hi;
  uint32_t nthreads, thread_id, iters;
  pthread_barrier_t* barrier;
  int64_t* cnt; /* per-thread arrays [ngroups] */
  int64_t* si;
  double* sf;
  uint64_t passed;
} OScanArg;

static void* o_scan_worker(void* argp) {
  OScanArg* a = (OScanArg*)argp;
  for (uint32_t it = 0; it < a->iters; ++it) {
    pthread_barrier_wait(a->barrier);
    memset(a->cnt, 0, sizeof(int64_t) * a->ngroups);
    memset(a->si, 0, sizeof(int64_t) * a->ngroups);
    memset(a->sf, 0, sizeof(double) * a->ngroups);
    a->passed = 0;
    const uint64_t per = (a->rows + a->nthreads - 1) / a->nthreads;
    const uint64_t r0 = (uint64_t)a->thread_id * per;
    const uint64_t r1 = r0 + per < a->rows ? r0 + per : a->rows;
    for (uint64_t r = r0; r < r1; ++r) {
      const int64_t x = a->v1[r];
      int ok;
      switch (a->pred_op) {
        case 1: ok = x < a->lo; break;
        case 2: ok = x >= a->lo; break;
        case 3: ok = x >= a->lo && x <= a->hi; break;
        default: ok = 1; break;
      }
      if (!ok) continue;
      ++a->passed;
      const uint32_t g = (uint32_t)a->keys[r];
      a->cnt[g] += 1;
      a->si[g] += x;
      a->sf[g] += (double)a->v2[r];
    }
    pthread_barrier_wait(a->barrier);
  }
  return NULL;
}

int o_scan_agg_mt(const int64_t* keys, const int64_t* v1, const float* v2,
                  uint64_t rows, uint32_t ngroups, int pred_op, int64_t lo,
                  int64_t hi, uint32_t nthreads, uint32_t iters,
                  int64_t* out_count, int64_t* out_sum_i64,
                  double* out_sum_f64, uint64_t* rows_passed) {
  if (nthreads == 0) nthreads = 1;
  if (iters == 0) iters = 1;
  OScanArg* args = (OScanArg*)calloc(nthreads, sizeof(OScanArg));
  pthread_t* th = (pthread_t*)malloc(sizeof(pthread_t) * nthreads);
  int64_t* cnt = (int64_t*)malloc(8ull * ngroups * nthreads);
  int64_t* si = (int64_t*)malloc(8ull * ngroups * nthreads);
  double* sf = (double*)malloc(8ull * ngroups * nthreads);
  pthread_barrier_t barrier;
  pthread_barrier_init(&barrier, NULL, nthreads);
  for (uint32_t t = 0; t < nthreads; ++t) {
    args[t] = (OScanArg){keys, v1, v2, rows, ngroups, pred_op, lo, hi,
                         nthreads, t, iters, &barrier,
                         cnt + (uint64_t)t * ngroups,
                         si + (uint64_t)t * ngroups,
                         sf + (uint64_t)t * ngroups, 0};
    pthread_create(&th[t], NULL, o_scan_worker, &args[t]);
  }
  memset(out_count, 0, 8ull * ngroups);
  memset(out_sum_i64, 0, 8ull * ngroups);
  memset(out_sum_f64, 0, 8ull * ngroups);
  uint64_t passed = 0;
  for (uint32_t t = 0; t < nthreads; ++t) {
    pthread_join(th[t], NULL);
    passed += args[t].passed;
    for (uint32_t g = 0; g < ngroups; ++g) {
      out_count[g] += args[t].cnt[g];
      out_sum_i64[g] += args[t].si[g];
      out_sum_f64[g] += args[t].sf[g];
    }
  }
  *rows_passed = passed;
  pthread_barrier_destroy(&barrier);
  free(args);
  free(th);
  free(cnt);
  free(si);
  free(sf);
  return 0;
}

int o_scan_agg(const int64_t* keys, const int64_t* v1, const float* v2,
               uint64_t rows, uint32_t ngroups, int pred_op, int64_t lo,
               int64_t hi, int64_t* out_count, int64_t* out_sum_i64,
               double* out_sum_f64, uint64_t* rows_passed) {
  memset(out_count, 0, sizeof(int64_t) * ngroups);
  memset(out_sum_i64, 0, sizeof(int64_t) * ngroups);
  memset(out_sum_f64, 0, sizeof(double) * ngroups);
  uint64_t passed = 0;
  for (uint64_t r = 0; r < rows; ++r) {
    const int64_t x = v1[r];
    int ok;
    switch (pred_op) {
      case 1: ok = x < lo; break;             /* LT */
      case 2: ok = x >= lo; break;            /* GE */
      case 3: ok = x >= lo && x <= hi; break; /* BETWEEN */
      default: ok = 1; break;
    }
    if (!ok) continue;
    ++passed;
    const uint32_t g = (uint32_t)keys[r];
    out_count[g] += 1;
    out_sum_i64[g] += x;
    out_sum_f64[g] += (double)v2[r];
  }
  *rows_passed = passed;
  return 0;
}
