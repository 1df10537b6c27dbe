// sdb_internal.h — shared internal context layout between sdb_gpu.hip and
// sdb_scan.hip (not part of the public C ABI).
#pragma once
#include <hip/hip_runtime.h>
#include "../../../include/sdb_gpu.h"

struct TermDev {
  uint64_t desc_begin;
  uint64_t desc_end;
  uint64_t payload_begin;
  float num;  // boost*(k+1)*idf  (Bm25Score::num, bm25.cpp:225)
  float nc;   // norm_const = k(1-b)
  float nl;   // norm_length = k*b/avgDL
};

struct SdbGpuCtx {
  int device;
  hipStream_t stream;
  // cached dense-scan workspace: the dense scan_agg hot path is
  // allocation-free across calls (grown on demand, freed with the ctx)
  unsigned long long* d_scan_out;
  uint64_t scan_out_cap;  // slots
  unsigned long long* d_scan_passed;
  SdbScoreDoc* d_cands;
  uint32_t* d_cand_count;
  unsigned long long* d_total_matches;
  uint32_t* d_gthresh;  // float bits
  uint32_t* d_ghist;    // global 256-bin score histogram (threshold tightening)
  unsigned long long* d_buckets;  // hybrid bucket aggregates [2*128]
  TermDev* d_terms;
  uint32_t* d_overflow;
  uint32_t* h_counts;  // pinned: [cand_count, overflow, final_bin]
  unsigned long long* h_matches;
  hipEvent_t ev_a, ev_b;   // bracket the window kernels of one execute call
  // pipelined-batch state (sdb_gpu_execute_topk_batch): second query-state
  // set + dedicated copy stream + pinned readback
  hipStream_t copy_stream;
  SdbScoreDoc* d_cands2;
  uint32_t* d_ghist2;
  unsigned char* d_qmisc;   // 2 x 64 B: {gthresh u32, pad, cand_count u32,
                            //  overflow u32, total u64} per set
  unsigned char* h_qmisc;   // pinned mirror
  TermDev* h_terms_pin;     // pinned term-table staging (SDB_TERM_SLOTS
                            // x SDB_MAX_TERMS): pageable-source async
                            // copies block the host until the stream
                            // drains, which serialized the batch pipeline
  SdbScoreDoc* h_cands_pin; // pinned candidate staging (SDB_PIN_CANDS)
  hipEvent_t ev_q[2];
  double last_kernel_ms;   // read back via sdb_gpu_last_kernel_ms
  // breakdown of the last execute_topk (sdb_gpu_last_stats)
  unsigned last_ncand;
  float last_gtau;
  double last_readback_ms;
  double last_select_ms;
};
