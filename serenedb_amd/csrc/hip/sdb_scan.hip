// sdb_scan.hip — columnar scan -> predicate -> hash-group-by on MI355X.
//
// Replaces the reference's FullScanner::Scan + ColFilterChain predicate
// narrowing (server/connector/full_scanner.h:40-90,
// index/table_filter_iterator.hpp:104-227) fused with the consumer the
// reference delegates to external DuckDB (PhysicalHashAggregate, un-vendored;
// result-level parity per SURVEY.md §8c).
//
// MI355X design: pure HBM-bandwidth work (no MFMA). Columns are either
// dense device-resident arrays or FoR/bitpack row groups (this repo's own
// codec, sdb_host.cpp: per-group frame-of-reference base + horizontal
// bitpack + min/max zonemap — the reference's block codecs live in the
// un-vendored DuckDB fork, so parity is at result level). The kernel walks
// row GROUPS grid-stride: one zonemap check per group can skip the whole
// group (DeadUntil analogue, full_scanner.h:61-71); within a group,
// threads decode packed values in registers (coalesced u32 payload reads),
// evaluate predicates, and accumulate into LDS per-group-key slots (dense
// keys = perfect hash); one device-atomic flush per block at kernel end.
// COUNT/SUM(i64) exact (wrap-around); SUM(f32) in f64 (atomic order
// nondeterministic, like the reference's thread-order-dependent fp
// aggregation; parity vs the oracle's sequential sum at ~1e-12 relative).

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <vector>

#include "../../../include/sdb_gpu.h"
#include "sdb_internal.h"

#ifndef SCAN_NTHREADS
#define SCAN_NTHREADS 1024u  // swept on-box: 1024t/2048b beats 256t/4096b;
#endif                       // with 4 rows/thread: 5.17 TB/s at 1B rows
#ifndef SCAN_MAXB
#define SCAN_MAXB 2048u
#endif
#define SCAN_MAX_GROUPS 2048u
#define SCAN_MAX_AGGS 8u
#define SCAN_MAX_PREDS 4u

#define HIP_CHECK(x)                                   \
  do {                                                 \
    hipError_t _e = (x);                               \
    if (_e == hipErrorNoDevice) return SDB_ERR_NO_GPU; \
    if (_e != hipSuccess) return SDB_ERR_HIP;          \
  } while (0)

// mirrors the host codec (sdb_host.cpp sdb_host_encode_col_i64)
struct SdbColHeaderDev {
  uint64_t magic;
  uint64_t rows;
  uint32_t group_rows;
  uint32_t ngroups;
  uint64_t off_desc;
  uint64_t off_payload;
  uint64_t size;
};
struct SdbColGroupDescDev {
  int64_t base;
  int64_t vmin;
  int64_t vmax;
  uint64_t word_off;
  uint16_t width;
  uint16_t pad[3];
};
#define SDB_COL_MAGIC_DEV 0x31304C4F43424453ull

struct ColRef {
  const void* data;                  // raw array, or FoR payload (u32*)
  const SdbColGroupDescDev* desc;    // FoR group table (null = raw)
};

#define SDB_MAX_STRCOLS 4
#define SDB_STRLIT_MAX 63  // literal bytes per side in strpred_mask

struct SdbGpuTable {
  void* cols[16];        // raw device array or whole FoR blob
  SdbColType types[16];
  ColRef refs[16];       // device pointers into cols[] allocations
  uint16_t max_w[16];    // FoR cols: max bit width over all groups
  uint64_t paywords[16]; // FoR cols: total payload u32 words (for clamped
                         // slack loads in the staged kernel)
  // i64 column value range, known for every FoR column (zonemaps) and for
  // every raw i64 column (one load-time GPU reduction): lets scan_agg
  // prove a group-key column fits [0, ngroups) instead of letting a stray
  // key scribble past the LDS accumulators (ADVICE r1 / VERDICT weak #4)
  int64_t col_min[16], col_max[16];
  uint8_t has_minmax[16];
  unsigned long long* valid[16];  // device validity bitmaps or null
  uint32_t ncols;
  uint64_t rows;
  uint32_t group_rows;   // shared by every FoR column (0 if none)
  // raw variable-width string columns (include/sdb_gpu.h attach_strcol):
  // per slot offsets[rows+1] + byte blob + the last strpred_mask result
  uint64_t* str_off[SDB_MAX_STRCOLS];
  uint8_t* str_blob[SDB_MAX_STRCOLS];
  uint64_t str_blob_len[SDB_MAX_STRCOLS];
  unsigned long long* str_mask[SDB_MAX_STRCOLS];
  uint8_t str_mask_set[SDB_MAX_STRCOLS];
  // FSST-style compressed slots: codes 0..nsym-1 expand to 1..8-byte
  // symbols, 255 escapes the next literal byte. The table is small and
  // travels by kernel argument (staged to LDS in the kernel).
  uint8_t str_fsst[SDB_MAX_STRCOLS];
  uint16_t str_nsym[SDB_MAX_STRCOLS];
  uint16_t str_symoff[SDB_MAX_STRCOLS][256];   // byte offsets, nsym+1 used
  uint8_t str_syms[SDB_MAX_STRCOLS][2048];
};

// load-time min/max reduction over a raw i64 column. Signed order via the
// sign-flip trick so u64 atomics suffice on every ROCm.
__global__ void col_minmax_kernel(const long long* __restrict__ v, uint64_t n,
                                  unsigned long long* __restrict__ mm) {
  const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  long long lmin = 0x7FFFFFFFFFFFFFFFll, lmax = 0x8000000000000000ll;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const long long x = v[i];
    lmin = x < lmin ? x : lmin;
    lmax = x > lmax ? x : lmax;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    const long long a = __shfl_down(lmin, off, 64);
    const long long b = __shfl_down(lmax, off, 64);
    lmin = a < lmin ? a : lmin;
    lmax = b > lmax ? b : lmax;
  }
  if ((threadIdx.x & 63) == 0) {
    atomicMin(&mm[0], (unsigned long long)lmin ^ 0x8000000000000000ull);
    atomicMax(&mm[1], (unsigned long long)lmax ^ 0x8000000000000000ull);
  }
}

struct ScanArgs {
  ColRef keys;
  uint64_t rows;
  uint32_t group_rows;  // row-group tiling (also used for raw-only tables)
  uint32_t ngroups;
  uint32_t naggs;
  uint32_t npreds;
  ColRef pred_col[SCAN_MAX_PREDS];
  int pred_op[SCAN_MAX_PREDS];
  int pred_isf32[SCAN_MAX_PREDS];  // f32 predicate columns
  int64_t pred_lo[SCAN_MAX_PREDS];
  int64_t pred_hi[SCAN_MAX_PREDS];
  float pred_flo[SCAN_MAX_PREDS], pred_fhi[SCAN_MAX_PREDS];
  const unsigned long long* pred_valid[SCAN_MAX_PREDS];  // null = all valid
  ColRef agg_col[SCAN_MAX_AGGS];
  int agg_op[SCAN_MAX_AGGS];
  int agg_src[SCAN_MAX_AGGS];  // 0=own col_read, 1+p=pred p's value, 9=key
  const unsigned long long* agg_valid[SCAN_MAX_AGGS];  // SUM skips nulls
  unsigned long long* out;
  unsigned long long* rows_passed;
  // --- staged-FoR variant only (scan_agg_staged_kernel) ---
  uint32_t chunk_rows;  // LDS staging chunk (multiple of 32)
  uint32_t nstage;      // distinct FoR cols staged (<=3: keys, pred0, pred1)
  int key_st;           // stage slot of the key column, -1 = unstaged
  int pred_st[2];       // stage slot of pred 0/1's column, -1 = unstaged
  const SdbColGroupDescDev* st_desc[3];
  const uint32_t* st_pay[3];
  uint64_t st_paywords[3];  // clamp for slack loads past the chunk
  uint32_t st_lds_off[3];   // u32 offset of each stage buffer in smem
  // optional f32 agg-column staging: the scattered per-passing-row 4 B
  // gather becomes one coalesced burst per chunk. Reads the whole column
  // (vs ~81% of its lines via the gather at 10% selectivity), so it can
  // lose on very low-pass non-dead groups; SDB_SCAN_NOF32STAGE opts out.
  const float* f32_pay;     // null = no f32 staging
  uint32_t f32_lds_off;
};

// predicate evaluation — the SdbPredOp set mirrors the reference's pushed
// table filters (index/table_filter_iterator.hpp:104-227: typed compares;
// EQ added round 2 per VERDICT #7). NaN floats fail every compare (SQL
// semantics of the reference's filter chain).
__device__ __forceinline__ bool pred_eval_i(int op, int64_t x, int64_t lo,
                                            int64_t hi) {
  switch (op) {
    case SDB_PRED_LT: return x < lo;
    case SDB_PRED_GE: return x >= lo;
    case SDB_PRED_BETWEEN: return (x >= lo) & (x <= hi);
    case SDB_PRED_EQ: return x == lo;
    default: return true;
  }
}
__device__ __forceinline__ bool valid_at(const unsigned long long* v,
                                         uint64_t r) {
  return !v || ((v[r >> 6] >> (r & 63u)) & 1ull);
}
// predicate with SQL three-valued logic over an optional validity plane:
// comparisons fail on NULL; ISNULL/NOTNULL evaluate the plane alone
__device__ __forceinline__ bool pred_eval_iv(
  int op, const unsigned long long* vb, uint64_t r, int64_t x, int64_t lo,
  int64_t hi) {
  if (op == SDB_PRED_ISNULL) return vb && !valid_at(vb, r);
  if (op == SDB_PRED_NOTNULL) return valid_at(vb, r);
  return valid_at(vb, r) && pred_eval_i(op, x, lo, hi);
}
__device__ __forceinline__ bool pred_eval_f(int op, float x, float lo,
                                            float hi) {
  switch (op) {
    case SDB_PRED_LT: return x < lo;
    case SDB_PRED_GE: return x >= lo;
    case SDB_PRED_BETWEEN: return (x >= lo) & (x <= hi);
    case SDB_PRED_EQ: return x == lo;
    default: return true;
  }
}

__device__ __forceinline__ bool pred_eval_fv(
  int op, const unsigned long long* vb, uint64_t r, float x, float lo,
  float hi) {
  if (op == SDB_PRED_ISNULL) return vb && !valid_at(vb, r);
  if (op == SDB_PRED_NOTNULL) return valid_at(vb, r);
  return valid_at(vb, r) && pred_eval_f(op, x, lo, hi);
}

// branchless funnel-shift extraction: value = bits [bit, bit+width) of the
// packed stream = v_alignbit_b32 of the two covering words (width <= 32 so
// one alignbit covers any in-word offset), masked. The +1 word read is
// unconditional; table_load over-allocates 4 pad bytes so the final value's
// slack read stays in bounds (its bits are masked out).
__device__ __forceinline__ uint32_t bits_at(const uint32_t* w, uint64_t bit) {
  const uint64_t wi = bit >> 5;
  return __builtin_amdgcn_alignbit(w[wi + 1], w[wi], (uint32_t)bit & 31u);
}

// desc fields are read per call from desc[rg]: the address is wave-uniform
// so the loads scalarize/broadcast from cache (a register-cached desc array
// indexed by a runtime agg index would spill to scratch — guide rule 20:
// that variant measured 2x slower than the raw path)
__device__ __forceinline__ int64_t col_read(const ColRef& c, uint32_t rg,
                                            uint64_t r0, uint64_t r) {
  if (!c.desc) return ((const long long*)c.data)[r];
  const SdbColGroupDescDev& d = c.desc[rg];
  if (d.width == 0) return d.base;
  const uint32_t* w = (const uint32_t*)c.data + d.word_off;
  const uint32_t mask =
    d.width >= 32 ? 0xFFFFFFFFu : ((1u << d.width) - 1u);
  return d.base + (int64_t)(bits_at(w, (r - r0) * d.width) & mask);
}

// paired FoR extraction: values r and r+1
__device__ __forceinline__ void col_read2(const ColRef& c, uint32_t rg,
                                          uint64_t r0, uint64_t r,
                                          int64_t& x0, int64_t& x1) {
  if (!c.desc) {
    longlong2 x;
    __builtin_memcpy(&x, &((const long long*)c.data)[r], 16);
    x0 = x.x;
    x1 = x.y;
    return;
  }
  const SdbColGroupDescDev& d = c.desc[rg];
  if (d.width == 0) {
    x0 = x1 = d.base;
    return;
  }
  const uint32_t* w = (const uint32_t*)c.data + d.word_off;
  const uint64_t bit0 = (r - r0) * d.width;
  const uint32_t mask =
    d.width >= 32 ? 0xFFFFFFFFu : ((1u << d.width) - 1u);
  x0 = d.base + (int64_t)(bits_at(w, bit0) & mask);
  x1 = d.base + (int64_t)(bits_at(w, bit0 + d.width) & mask);
}

// staged analogue of col_read2: the chunk's packed words sit in LDS (sw),
// bit offsets are chunk-relative (lr = row - chunk_start; chunks start on a
// 32-row multiple so the chunk's first bit is word-aligned). Desc fields are
// read per call — the address is wave-uniform so the loads scalarize, same
// rationale as col_read.
__device__ __forceinline__ void col_read2_lds(const SdbColGroupDescDev* desc,
                                              const uint32_t* sw, uint32_t rg,
                                              uint32_t lr, int64_t& x0,
                                              int64_t& x1) {
  const SdbColGroupDescDev& d = desc[rg];
  if (d.width == 0) {
    x0 = x1 = d.base;
    return;
  }
  const uint64_t bit0 = (uint64_t)lr * d.width;
  const uint32_t mask =
    d.width >= 32 ? 0xFFFFFFFFu : ((1u << d.width) - 1u);
  x0 = d.base + (int64_t)(bits_at(sw, bit0) & mask);
  x1 = d.base + (int64_t)(bits_at(sw, bit0 + d.width) & mask);
}

// ASHAPE: compile-time aggregate-shape dispatch (DuckDB-style operator
// specialization). 0 = generic runtime loop; 1 = the common 3-agg shape
// [COUNT(*), SUM(i64), SUM(f32->f64)] with the per-row per-agg op/src
// switches straight-lined (the FoR walker is instruction-issue-bound —
// tools/ROUND2_NOTES.md). Any other shape falls back to ASHAPE=0.
template <int RAW, int ASHAPE>
__launch_bounds__(SCAN_NTHREADS) __global__ void scan_agg_kernel(ScanArgs a) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned long long* acc = (unsigned long long*)smem;  // ngroups*naggs
  const uint32_t nslots = a.ngroups * a.naggs;
  for (uint32_t i = threadIdx.x; i < nslots; i += SCAN_NTHREADS) acc[i] = 0;
  __syncthreads();

  if (RAW) {
    // raw dense columns: 4 rows/thread with 2x16-byte loads per column
    // (measured +9% over 2 rows/thread: 1.79 -> 1.64 ms per 400M rows)
    const uint64_t quad_stride = (uint64_t)gridDim.x * SCAN_NTHREADS * 4u;
    unsigned long long my_passed = 0;
    const uint64_t rows4 = a.rows & ~3ull;
    for (uint64_t r =
           ((uint64_t)blockIdx.x * SCAN_NTHREADS + threadIdx.x) * 4u;
         r < rows4; r += quad_stride) {
      bool okv[4] = {true, true, true, true};
      for (uint32_t p = 0; p < a.npreds; ++p) {
        if (a.pred_isf32[p]) {
          float4 f;
          __builtin_memcpy(&f, &((const float*)a.pred_col[p].data)[r], 16);
          const float fs[4] = {f.x, f.y, f.z, f.w};
#pragma unroll
          for (int e = 0; e < 4; ++e)
            okv[e] &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r + e,
                                   fs[e], a.pred_flo[p], a.pred_fhi[p]);
          continue;
        }
        longlong2 x, y;
        __builtin_memcpy(&x, &((const long long*)a.pred_col[p].data)[r], 16);
        __builtin_memcpy(&y, &((const long long*)a.pred_col[p].data)[r + 2],
                         16);
        const int64_t xs[4] = {x.x, x.y, y.x, y.y};
#pragma unroll
        for (int e = 0; e < 4; ++e)
          okv[e] &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r + e,
                                 xs[e], a.pred_lo[p], a.pred_hi[p]);
      }
      if (!okv[0] && !okv[1] && !okv[2] && !okv[3]) continue;
      longlong2 k0, k1;
      __builtin_memcpy(&k0, &((const long long*)a.keys.data)[r], 16);
      __builtin_memcpy(&k1, &((const long long*)a.keys.data)[r + 2], 16);
      const int64_t ks[4] = {k0.x, k0.y, k1.x, k1.y};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        if (!okv[e]) continue;
        ++my_passed;
        const uint32_t g = (uint32_t)ks[e];
        if (ASHAPE == 1) {  // straight-lined 3-agg shape (dead if generic)
          unsigned long long* s3 = &acc[g * 3u];
          atomicAdd(s3, 1ull);
          if (valid_at(a.agg_valid[1], r + e))
            atomicAdd(s3 + 1, (unsigned long long)((const long long*)
                                                     a.agg_col[1]
                                                       .data)[r + e]);
          if (valid_at(a.agg_valid[2], r + e))
            atomicAdd((double*)(s3 + 2),
                      (double)((const float*)a.agg_col[2].data)[r + e]);
          continue;
        }
        for (uint32_t q = 0; q < a.naggs; ++q) {
          unsigned long long* slot = &acc[g * a.naggs + q];
          if (a.agg_op[q] != SDB_AGG_COUNT &&
              !valid_at(a.agg_valid[q], r + e))
            continue;  // SUM skips NULL values; COUNT(*) counts the row
          switch (a.agg_op[q]) {
            case SDB_AGG_COUNT:
              atomicAdd(slot, 1ull);
              break;
            case SDB_AGG_SUM_I64:
              atomicAdd(slot, (unsigned long long)((const long long*)
                                                     a.agg_col[q]
                                                       .data)[r + e]);
              break;
            case SDB_AGG_SUM_F64:
              atomicAdd((double*)slot,
                        (double)((const float*)a.agg_col[q].data)[r + e]);
              break;
          }
        }
      }
    }
    if ((a.rows & 3ull) && blockIdx.x == 0 && threadIdx.x == 0)
      for (uint64_t r = rows4; r < a.rows; ++r) {
      bool ok = true;
      for (uint32_t p = 0; p < a.npreds; ++p) {
        if (a.pred_isf32[p])
          ok &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r,
                             ((const float*)a.pred_col[p].data)[r],
                             a.pred_flo[p], a.pred_fhi[p]);
        else
          ok &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r,
                             ((const long long*)a.pred_col[p].data)[r],
                             a.pred_lo[p], a.pred_hi[p]);
      }
      if (ok) {
        ++my_passed;
        const uint32_t g = (uint32_t)((const long long*)a.keys.data)[r];
        for (uint32_t q = 0; q < a.naggs; ++q) {
          unsigned long long* slot = &acc[g * a.naggs + q];
          if (a.agg_op[q] != SDB_AGG_COUNT &&
              !valid_at(a.agg_valid[q], r))
            continue;
          switch (a.agg_op[q]) {
            case SDB_AGG_COUNT: atomicAdd(slot, 1ull); break;
            case SDB_AGG_SUM_I64:
              atomicAdd(slot, (unsigned long long)((const long long*)
                                                     a.agg_col[q].data)[r]);
              break;
            case SDB_AGG_SUM_F64:
              atomicAdd((double*)slot,
                        (double)((const float*)a.agg_col[q].data)[r]);
              break;
          }
        }
      }
    }
    unsigned long long wp = my_passed;
#pragma unroll
    for (int off = 32; off; off >>= 1) wp += __shfl_down(wp, off, 64);
    if ((threadIdx.x & 63) == 0 && wp) atomicAdd(a.rows_passed, wp);
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < nslots; i += SCAN_NTHREADS) {
      const uint32_t q = i % a.naggs;
      if (a.agg_op[q] == SDB_AGG_SUM_F64) {
        double v;
        __builtin_memcpy(&v, &acc[i], 8);
        if (v != 0.0) atomicAdd((double*)&a.out[i], v);
      } else if (acc[i]) {
        atomicAdd(&a.out[i], acc[i]);
      }
    }
    return;
  }

  const uint32_t n_rowgroups =
    (uint32_t)((a.rows + a.group_rows - 1) / a.group_rows);
  unsigned long long my_passed = 0;

  for (uint32_t rg = blockIdx.x; rg < n_rowgroups; rg += gridDim.x) {
    const uint64_t r0 = (uint64_t)rg * a.group_rows;
    const uint64_t r1 = min(a.rows, r0 + a.group_rows);
    // zonemap skip (DeadUntil analogue)
    bool dead = false;
    for (uint32_t p = 0; p < a.npreds; ++p) {
      if (a.pred_col[p].desc) {
        const SdbColGroupDescDev& d = a.pred_col[p].desc[rg];
        switch (a.pred_op[p]) {
          case SDB_PRED_LT: dead |= d.vmin >= a.pred_lo[p]; break;
          case SDB_PRED_GE: dead |= d.vmax < a.pred_lo[p]; break;
          case SDB_PRED_BETWEEN:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_hi[p]);
            break;
          case SDB_PRED_EQ:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_lo[p]);
            break;
          default: break;
        }
      }
    }
    if (dead) continue;

    // 2 rows per thread, paired word extraction; odd tail row by thread 0
    const uint64_t glen = r1 - r0;
    const uint64_t gpairs = glen >> 1;
    for (uint64_t pr = threadIdx.x; pr < gpairs; pr += SCAN_NTHREADS) {
      const uint64_t r = r0 + 2 * pr;
      bool okv[2] = {true, true};
      int64_t pva[2] = {0, 0}, pvb[2] = {0, 0};
#pragma unroll
      for (uint32_t p = 0; p < 2; ++p) {  // fast path: first two preds
        if (p >= a.npreds) break;
        if (a.pred_isf32[p]) {
          const float* fc = (const float*)a.pred_col[p].data;
#pragma unroll
          for (int e = 0; e < 2; ++e)
            okv[e] &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r + e,
                                   fc[r + e], a.pred_flo[p],
                                   a.pred_fhi[p]);
          continue;
        }
        int64_t x0, x1;
        col_read2(a.pred_col[p], rg, r0, r, x0, x1);
        if (p == 0) { pva[0] = x0; pva[1] = x1; }
        else { pvb[0] = x0; pvb[1] = x1; }
        const int64_t xs[2] = {x0, x1};
#pragma unroll
        for (int e = 0; e < 2; ++e)
          okv[e] &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r + e,
                                 xs[e], a.pred_lo[p], a.pred_hi[p]);
      }
      for (uint32_t p = 2; p < a.npreds; ++p) {  // rare: >2 predicates
#pragma unroll
        for (int e = 0; e < 2; ++e) {
          if (a.pred_isf32[p])
            okv[e] &= pred_eval_fv(
              a.pred_op[p], a.pred_valid[p], r + e,
              ((const float*)a.pred_col[p].data)[r + e], a.pred_flo[p],
              a.pred_fhi[p]);
          else
            okv[e] &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r + e,
                                   col_read(a.pred_col[p], rg, r0, r + e),
                                   a.pred_lo[p], a.pred_hi[p]);
        }
      }
      if (!okv[0] && !okv[1]) continue;
      int64_t k0, k1;
      col_read2(a.keys, rg, r0, r, k0, k1);
      const int64_t ks[2] = {k0, k1};
#pragma unroll
      for (int e = 0; e < 2; ++e) {
        if (!okv[e]) continue;
        ++my_passed;
        const uint32_t grp = (uint32_t)ks[e];
        if (ASHAPE == 1) {  // straight-lined 3-agg shape (dead if generic)
          unsigned long long* s3 = &acc[grp * 3u];
          atomicAdd(s3, 1ull);
          if (valid_at(a.agg_valid[1], r + e)) {
            int64_t x;
            switch (a.agg_src[1]) {
              case 1: x = pva[e]; break;
              case 2: x = pvb[e]; break;
              case 9: x = ks[e]; break;
              default: x = col_read(a.agg_col[1], rg, r0, r + e); break;
            }
            atomicAdd(s3 + 1, (unsigned long long)x);
          }
          if (valid_at(a.agg_valid[2], r + e))
            atomicAdd((double*)(s3 + 2),
                      (double)((const float*)a.agg_col[2].data)[r + e]);
          continue;
        }
        for (uint32_t q = 0; q < a.naggs; ++q) {
          unsigned long long* slot = &acc[grp * a.naggs + q];
          if (a.agg_op[q] != SDB_AGG_COUNT &&
              !valid_at(a.agg_valid[q], r + e))
            continue;
          switch (a.agg_op[q]) {
            case SDB_AGG_COUNT:
              atomicAdd(slot, 1ull);
              break;
            case SDB_AGG_SUM_I64: {
              int64_t x;
              switch (a.agg_src[q]) {  // decode-once dedup vs preds/key
                case 1: x = pva[e]; break;
                case 2: x = pvb[e]; break;
                case 9: x = ks[e]; break;
                default: x = col_read(a.agg_col[q], rg, r0, r + e); break;
              }
              atomicAdd(slot, (unsigned long long)x);
              break;
            }
            case SDB_AGG_SUM_F64:
              atomicAdd((double*)slot,
                        (double)((const float*)a.agg_col[q].data)[r + e]);
              break;
          }
        }
      }
    }
    if ((glen & 1ull) && threadIdx.x == 0) {
      const uint64_t r = r1 - 1;
      bool ok = true;
      int64_t pv0s = 0, pv1s = 0;
      for (uint32_t p = 0; p < a.npreds; ++p) {
        if (a.pred_isf32[p]) {
          ok &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r,
                             ((const float*)a.pred_col[p].data)[r],
                             a.pred_flo[p], a.pred_fhi[p]);
          continue;
        }
        const int64_t x = col_read(a.pred_col[p], rg, r0, r);
        if (p == 0) pv0s = x;
        else if (p == 1) pv1s = x;
        ok &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r, x,
                           a.pred_lo[p], a.pred_hi[p]);
      }
      if (ok) {
        ++my_passed;
        const uint32_t grp = (uint32_t)col_read(a.keys, rg, r0, r);
        if (ASHAPE == 1) {  // straight-lined 3-agg shape
          unsigned long long* s3 = &acc[grp * 3u];
          atomicAdd(s3, 1ull);
          if (valid_at(a.agg_valid[1], r)) {
            int64_t x;
            switch (a.agg_src[1]) {
              case 1: x = pv0s; break;
              case 2: x = pv1s; break;
              case 9: x = (int64_t)grp; break;
              default: x = col_read(a.agg_col[1], rg, r0, r); break;
            }
            atomicAdd(s3 + 1, (unsigned long long)x);
          }
          if (valid_at(a.agg_valid[2], r))
            atomicAdd((double*)(s3 + 2),
                      (double)((const float*)a.agg_col[2].data)[r]);
        } else
        for (uint32_t q = 0; q < a.naggs; ++q) {
          unsigned long long* slot = &acc[grp * a.naggs + q];
          if (a.agg_op[q] != SDB_AGG_COUNT &&
              !valid_at(a.agg_valid[q], r))
            continue;
          switch (a.agg_op[q]) {
            case SDB_AGG_COUNT: atomicAdd(slot, 1ull); break;
            case SDB_AGG_SUM_I64: {
              int64_t x;
              switch (a.agg_src[q]) {
                case 1: x = pv0s; break;
                case 2: x = pv1s; break;
                case 9: x = (int64_t)grp; break;
                default: x = col_read(a.agg_col[q], rg, r0, r); break;
              }
              atomicAdd(slot, (unsigned long long)x);
              break;
            }
            case SDB_AGG_SUM_F64:
              atomicAdd((double*)slot,
                        (double)((const float*)a.agg_col[q].data)[r]);
              break;
          }
        }
      }
    }
  }

  unsigned long long wp = my_passed;
#pragma unroll
  for (int off = 32; off; off >>= 1) wp += __shfl_down(wp, off, 64);
  if ((threadIdx.x & 63) == 0 && wp) atomicAdd(a.rows_passed, wp);
  __syncthreads();
  for (uint32_t i = threadIdx.x; i < nslots; i += SCAN_NTHREADS) {
    const uint32_t q = i % a.naggs;
    if (a.agg_op[q] == SDB_AGG_SUM_F64) {
      double v;
      __builtin_memcpy(&v, &acc[i], 8);
      if (v != 0.0) atomicAdd((double*)&a.out[i], v);
    } else if (acc[i]) {
      atomicAdd(&a.out[i], acc[i]);
    }
  }
}

// FoR walker with LDS payload staging: the plain walker (scan_agg_kernel<0>)
// is latency-bound on its per-pair global extract chains (3 dependent word
// loads per column per pair; measured 137G rows/s vs the raw path's 233G).
// Here each chunk's packed words for the key + first two predicate columns
// are staged cooperatively (one coalesced burst per column), and all pair
// extraction reads hit LDS. Zonemap skips, predicate fast path, and the
// agg_src decode-once dedup are identical to the unstaged walker.
template <int ASHAPE>
__launch_bounds__(SCAN_NTHREADS) __global__
void scan_agg_staged_kernel(ScanArgs a) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned long long* acc = (unsigned long long*)smem;  // ngroups*naggs
  uint32_t* sw = (uint32_t*)smem;  // stage buffers live past acc
  const uint32_t nslots = a.ngroups * a.naggs;
  for (uint32_t i = threadIdx.x; i < nslots; i += SCAN_NTHREADS) acc[i] = 0;
  __syncthreads();  // a 1-row group accumulates before any chunk barrier

  const uint32_t n_rowgroups =
    (uint32_t)((a.rows + a.group_rows - 1) / a.group_rows);
  unsigned long long my_passed = 0;

  for (uint32_t rg = blockIdx.x; rg < n_rowgroups; rg += gridDim.x) {
    const uint64_t r0 = (uint64_t)rg * a.group_rows;
    const uint64_t r1 = min(a.rows, r0 + a.group_rows);
    // zonemap skip (DeadUntil analogue)
    bool dead = false;
    for (uint32_t p = 0; p < a.npreds; ++p) {
      if (a.pred_col[p].desc) {
        const SdbColGroupDescDev& d = a.pred_col[p].desc[rg];
        switch (a.pred_op[p]) {
          case SDB_PRED_LT: dead |= d.vmin >= a.pred_lo[p]; break;
          case SDB_PRED_GE: dead |= d.vmax < a.pred_lo[p]; break;
          case SDB_PRED_BETWEEN:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_hi[p]);
            break;
          case SDB_PRED_EQ:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_lo[p]);
            break;
          default: break;
        }
      }
    }
    if (dead) continue;

    const uint64_t glen = r1 - r0;
    const uint64_t glen_even = glen & ~1ull;
    for (uint64_t c0 = 0; c0 < glen_even; c0 += a.chunk_rows) {
      const uint32_t clen =
        (uint32_t)min((uint64_t)a.chunk_rows, glen_even - c0);
      __syncthreads();  // prior chunk's extraction done before overwrite
      for (uint32_t s = 0; s < a.nstage; ++s) {
        const SdbColGroupDescDev& d = a.st_desc[s][rg];
        if (d.width == 0) continue;
        // chunk starts on a 32-row multiple => word-aligned bit offset
        const uint64_t wstart = d.word_off + ((c0 * d.width) >> 5);
        const uint32_t nw =
          (uint32_t)(((uint64_t)clen * d.width + 31) >> 5) + 2;
        const uint64_t limit = a.st_paywords[s] - 1;
        uint32_t* dst = sw + a.st_lds_off[s];
        const uint32_t* src = a.st_pay[s];
        for (uint32_t i = threadIdx.x; i < nw; i += SCAN_NTHREADS)
          dst[i] = src[min(wstart + i, limit)];
      }
      if (a.f32_pay) {
        const float* src = a.f32_pay + (r0 + c0);
        float* dst = (float*)(sw + a.f32_lds_off);
        for (uint32_t i = threadIdx.x; i < clen; i += SCAN_NTHREADS)
          dst[i] = src[i];
      }
      __syncthreads();
      const uint32_t cpairs = clen >> 1;
      for (uint32_t pr = threadIdx.x; pr < cpairs; pr += SCAN_NTHREADS) {
        const uint32_t lr = 2 * pr;       // chunk-relative row
        const uint64_t r = r0 + c0 + lr;  // absolute row
        bool okv[2] = {true, true};
        int64_t pva[2] = {0, 0}, pvb[2] = {0, 0};
#pragma unroll
        for (uint32_t p = 0; p < 2; ++p) {  // fast path: first two preds
          if (p >= a.npreds) break;
          if (a.pred_isf32[p]) {
            const float* fc = (const float*)a.pred_col[p].data;
#pragma unroll
            for (int e = 0; e < 2; ++e)
              okv[e] &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r + e,
                                     fc[r + e], a.pred_flo[p],
                                     a.pred_fhi[p]);
            continue;
          }
          int64_t x0, x1;
          if (a.pred_st[p] >= 0)
            col_read2_lds(a.st_desc[a.pred_st[p]],
                          sw + a.st_lds_off[a.pred_st[p]], rg, lr, x0, x1);
          else
            col_read2(a.pred_col[p], rg, r0, r, x0, x1);
          if (p == 0) { pva[0] = x0; pva[1] = x1; }
          else { pvb[0] = x0; pvb[1] = x1; }
          const int64_t xs[2] = {x0, x1};
#pragma unroll
          for (int e = 0; e < 2; ++e)
            okv[e] &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r + e,
                                   xs[e], a.pred_lo[p], a.pred_hi[p]);
        }
        for (uint32_t p = 2; p < a.npreds; ++p) {  // rare: >2 predicates
#pragma unroll
          for (int e = 0; e < 2; ++e) {
            if (a.pred_isf32[p])
              okv[e] &= pred_eval_fv(
                a.pred_op[p], a.pred_valid[p], r + e,
                ((const float*)a.pred_col[p].data)[r + e], a.pred_flo[p],
                a.pred_fhi[p]);
            else
              okv[e] &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r + e,
                                     col_read(a.pred_col[p], rg, r0,
                                              r + e),
                                     a.pred_lo[p], a.pred_hi[p]);
          }
        }
        if (!okv[0] && !okv[1]) continue;
        int64_t k0, k1;
        if (a.key_st >= 0)
          col_read2_lds(a.st_desc[a.key_st], sw + a.st_lds_off[a.key_st],
                        rg, lr, k0, k1);
        else
          col_read2(a.keys, rg, r0, r, k0, k1);
        const int64_t ks[2] = {k0, k1};
#pragma unroll
        for (int e = 0; e < 2; ++e) {
          if (!okv[e]) continue;
          ++my_passed;
          const uint32_t grp = (uint32_t)ks[e];
          if (ASHAPE == 1) {  // straight-lined 3-agg shape
            unsigned long long* s3 = &acc[grp * 3u];
            atomicAdd(s3, 1ull);
            if (valid_at(a.agg_valid[1], r + e)) {
              int64_t x;
              switch (a.agg_src[1]) {
                case 1: x = pva[e]; break;
                case 2: x = pvb[e]; break;
                case 9: x = ks[e]; break;
                default: x = col_read(a.agg_col[1], rg, r0, r + e); break;
              }
              atomicAdd(s3 + 1, (unsigned long long)x);
            }
            if (valid_at(a.agg_valid[2], r + e)) {
              const float* fc = (const float*)a.agg_col[2].data;
              const float fv = fc == a.f32_pay
                                 ? ((const float*)(sw +
                                                   a.f32_lds_off))[lr + e]
                                 : fc[r + e];
              atomicAdd((double*)(s3 + 2), (double)fv);
            }
            continue;
          }
          for (uint32_t q = 0; q < a.naggs; ++q) {
            unsigned long long* slot = &acc[grp * a.naggs + q];
            if (a.agg_op[q] != SDB_AGG_COUNT &&
                !valid_at(a.agg_valid[q], r + e))
              continue;
            switch (a.agg_op[q]) {
              case SDB_AGG_COUNT:
                atomicAdd(slot, 1ull);
                break;
              case SDB_AGG_SUM_I64: {
                int64_t x;
                switch (a.agg_src[q]) {  // decode-once dedup vs preds/key
                  case 1: x = pva[e]; break;
                  case 2: x = pvb[e]; break;
                  case 9: x = ks[e]; break;
                  default: x = col_read(a.agg_col[q], rg, r0, r + e); break;
                }
                atomicAdd(slot, (unsigned long long)x);
                break;
              }
              case SDB_AGG_SUM_F64: {
                const float* fc = (const float*)a.agg_col[q].data;
                const float fv = fc == a.f32_pay
                                   ? ((const float*)(sw +
                                                     a.f32_lds_off))[lr + e]
                                   : fc[r + e];
                atomicAdd((double*)slot, (double)fv);
                break;
              }
            }
          }
        }
      }
    }
    if ((glen & 1ull) && threadIdx.x == 0) {  // odd tail row: global reads
      const uint64_t r = r1 - 1;
      bool ok = true;
      int64_t pv0s = 0, pv1s = 0;
      for (uint32_t p = 0; p < a.npreds; ++p) {
        if (a.pred_isf32[p]) {
          ok &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r,
                             ((const float*)a.pred_col[p].data)[r],
                             a.pred_flo[p], a.pred_fhi[p]);
          continue;
        }
        const int64_t x = col_read(a.pred_col[p], rg, r0, r);
        if (p == 0) pv0s = x;
        else if (p == 1) pv1s = x;
        ok &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r, x,
                           a.pred_lo[p], a.pred_hi[p]);
      }
      if (ok) {
        ++my_passed;
        const uint32_t grp = (uint32_t)col_read(a.keys, rg, r0, r);
        if (ASHAPE == 1) {  // straight-lined 3-agg shape
          unsigned long long* s3 = &acc[grp * 3u];
          atomicAdd(s3, 1ull);
          if (valid_at(a.agg_valid[1], r)) {
            int64_t x;
            switch (a.agg_src[1]) {
              case 1: x = pv0s; break;
              case 2: x = pv1s; break;
              case 9: x = (int64_t)grp; break;
              default: x = col_read(a.agg_col[1], rg, r0, r); break;
            }
            atomicAdd(s3 + 1, (unsigned long long)x);
          }
          if (valid_at(a.agg_valid[2], r))
            atomicAdd((double*)(s3 + 2),
                      (double)((const float*)a.agg_col[2].data)[r]);
        } else
        for (uint32_t q = 0; q < a.naggs; ++q) {
          unsigned long long* slot = &acc[grp * a.naggs + q];
          if (a.agg_op[q] != SDB_AGG_COUNT &&
              !valid_at(a.agg_valid[q], r))
            continue;
          switch (a.agg_op[q]) {
            case SDB_AGG_COUNT: atomicAdd(slot, 1ull); break;
            case SDB_AGG_SUM_I64: {
              int64_t x;
              switch (a.agg_src[q]) {
                case 1: x = pv0s; break;
                case 2: x = pv1s; break;
                case 9: x = (int64_t)grp; break;
                default: x = col_read(a.agg_col[q], rg, r0, r); break;
              }
              atomicAdd(slot, (unsigned long long)x);
              break;
            }
            case SDB_AGG_SUM_F64:
              atomicAdd((double*)slot,
                        (double)((const float*)a.agg_col[q].data)[r]);
              break;
          }
        }
      }
    }
  }

  unsigned long long wp = my_passed;
#pragma unroll
  for (int off = 32; off; off >>= 1) wp += __shfl_down(wp, off, 64);
  if ((threadIdx.x & 63) == 0 && wp) atomicAdd(a.rows_passed, wp);
  __syncthreads();
  for (uint32_t i = threadIdx.x; i < nslots; i += SCAN_NTHREADS) {
    const uint32_t q = i % a.naggs;
    if (a.agg_op[q] == SDB_AGG_SUM_F64) {
      double v;
      __builtin_memcpy(&v, &acc[i], 8);
      if (v != 0.0) atomicAdd((double*)&a.out[i], v);
    } else if (acc[i]) {
      atomicAdd(&a.out[i], acc[i]);
    }
  }
}


// ---------------------------------------------------------------------------
// General hash aggregate (round-2): arbitrary i64 group keys.
// north_star: "hash-group-by with LDS-staged open-addressed buckets" — the
// consumer the reference hands to DuckDB's PhysicalHashAggregate
// (duckdb_search_full_scan.cpp:2141-2171) for keys the dense perfect-hash
// kernel cannot take. Per-workgroup open-addressed LDS table (linear
// probing, 64-bit key CAS), flushed once into a global open-addressed
// table; rows that miss the LDS probe bound go straight to the global
// table. Exact integer aggregates; SUM(f32) in f64 (atomic order
// nondeterministic, as the dense kernel).
// ---------------------------------------------------------------------------

#define SDB_HKEY_EMPTY 0xFFFFFFFFFFFFFFFFull /* key -1 uses neg_acc */

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  // splitmix64 finalizer
  x ^= x >> 30;
  x *= 0xbf58476d1ce4e5b9ull;
  x ^= x >> 27;
  x *= 0x94d049bb133111ebull;
  x ^= x >> 31;
  return x;
}

struct HashAggArgs {
  ColRef keys;
  uint64_t rows;
  uint32_t group_rows;
  uint32_t naggs;
  uint32_t npreds;
  ColRef pred_col[SCAN_MAX_PREDS];
  int pred_op[SCAN_MAX_PREDS];
  int pred_isf32[SCAN_MAX_PREDS];
  int64_t pred_lo[SCAN_MAX_PREDS];
  int64_t pred_hi[SCAN_MAX_PREDS];
  float pred_flo[SCAN_MAX_PREDS], pred_fhi[SCAN_MAX_PREDS];
  const unsigned long long* pred_valid[SCAN_MAX_PREDS];
  ColRef agg_col[SCAN_MAX_AGGS];
  int agg_op[SCAN_MAX_AGGS];
  const unsigned long long* agg_valid[SCAN_MAX_AGGS];
  // global open-addressed table (pow2 capacity, keys init SDB_HKEY_EMPTY)
  unsigned long long* gkeys;
  unsigned long long* gacc;  // [cap * naggs]
  uint32_t gcap_mask;
  uint32_t max_groups;
  uint32_t* ginserts;   // distinct-key counter (contract check)
  uint32_t* goverflow;  // probe bound exceeded
  unsigned long long* neg_acc;  // [naggs+1]: key == -1 accumulators + flag
  unsigned long long* rows_passed;
  uint32_t lds_slots;   // pow2 (0 = LDS staging disabled)
};

// probe/insert into the global table; returns slot or -1 (overflow).
__device__ __forceinline__ int64_t hash_global_upsert(
  const HashAggArgs& a, uint64_t k) {
  uint32_t h = (uint32_t)mix64(k) & a.gcap_mask;
  for (uint32_t i = 0; i <= a.gcap_mask; ++i) {
    const unsigned long long cur = __hip_atomic_load(
      &a.gkeys[h], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (cur == k) return (int64_t)h;
    if (cur == SDB_HKEY_EMPTY) {
      const unsigned long long prev =
        atomicCAS(&a.gkeys[h], SDB_HKEY_EMPTY, k);
      if (prev == SDB_HKEY_EMPTY) {
        atomicAdd(a.ginserts, 1u);
        return (int64_t)h;
      }
      if (prev == k) return (int64_t)h;
    }
    h = (h + 1) & a.gcap_mask;
  }
  atomicExch(a.goverflow, 1u);
  return -1;
}

__device__ __forceinline__ void hash_acc_add(const HashAggArgs& a,
                                             unsigned long long* acc,
                                             uint32_t q, uint32_t rg,
                                             uint64_t r0, uint64_t r,
                                             bool global_scope) {
  if (a.agg_op[q] != SDB_AGG_COUNT && !valid_at(a.agg_valid[q], r))
    return;  // SUM skips NULL values; COUNT(*) counts the row
  switch (a.agg_op[q]) {
    case SDB_AGG_COUNT:
      atomicAdd(&acc[q], 1ull);
      break;
    case SDB_AGG_SUM_I64:
      atomicAdd(&acc[q],
                (unsigned long long)col_read(a.agg_col[q], rg, r0, r));
      break;
    case SDB_AGG_SUM_F64:
      atomicAdd((double*)&acc[q],
                (double)((const float*)a.agg_col[q].data)[r]);
      break;
  }
  (void)global_scope;
}

__launch_bounds__(SCAN_NTHREADS) __global__
void scan_agg_hash_kernel(HashAggArgs a) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned long long* lkey = (unsigned long long*)smem;     // lds_slots
  unsigned long long* lacc = lkey + a.lds_slots;            // slots*naggs
  const uint32_t S_mask = a.lds_slots ? a.lds_slots - 1 : 0;
  for (uint32_t i = threadIdx.x; i < a.lds_slots * (1 + a.naggs);
       i += SCAN_NTHREADS)
    lkey[i] = i < a.lds_slots ? SDB_HKEY_EMPTY : 0ull;
  __syncthreads();

  const uint32_t n_rowgroups =
    (uint32_t)((a.rows + a.group_rows - 1) / a.group_rows);
  unsigned long long my_passed = 0;

  for (uint32_t rg = blockIdx.x; rg < n_rowgroups; rg += gridDim.x) {
    const uint64_t r0 = (uint64_t)rg * a.group_rows;
    const uint64_t r1 = min(a.rows, r0 + a.group_rows);
    bool dead = false;
    for (uint32_t p = 0; p < a.npreds; ++p) {
      if (a.pred_col[p].desc && !a.pred_isf32[p]) {
        const SdbColGroupDescDev& d = a.pred_col[p].desc[rg];
        switch (a.pred_op[p]) {
          case SDB_PRED_LT: dead |= d.vmin >= a.pred_lo[p]; break;
          case SDB_PRED_GE: dead |= d.vmax < a.pred_lo[p]; break;
          case SDB_PRED_BETWEEN:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_hi[p]);
            break;
          case SDB_PRED_EQ:
            dead |= (d.vmax < a.pred_lo[p]) | (d.vmin > a.pred_lo[p]);
            break;
          default: break;
        }
      }
    }
    if (dead) continue;

    for (uint64_t r = r0 + threadIdx.x; r < r1; r += SCAN_NTHREADS) {
      bool ok = true;
      for (uint32_t p = 0; p < a.npreds; ++p) {
        if (a.pred_isf32[p])
          ok &= pred_eval_fv(a.pred_op[p], a.pred_valid[p], r,
                             ((const float*)a.pred_col[p].data)[r],
                             a.pred_flo[p], a.pred_fhi[p]);
        else
          ok &= pred_eval_iv(a.pred_op[p], a.pred_valid[p], r,
                             col_read(a.pred_col[p], rg, r0, r),
                             a.pred_lo[p], a.pred_hi[p]);
      }
      if (!ok) continue;
      ++my_passed;
      const uint64_t k = (uint64_t)col_read(a.keys, rg, r0, r);
      if (k == SDB_HKEY_EMPTY) {  // actual key -1: dedicated accumulators
        if (!__hip_atomic_load(&a.neg_acc[a.naggs], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT))
          atomicExch(&a.neg_acc[a.naggs], 1ull);
        for (uint32_t q = 0; q < a.naggs; ++q)
          hash_acc_add(a, a.neg_acc, q, rg, r0, r, true);
        continue;
      }
      // LDS open-addressed probe (bounded), then global fallback
      int slot = -1;
      if (a.lds_slots) {
        uint32_t h = (uint32_t)mix64(k) & S_mask;
        for (uint32_t i = 0; i < 32; ++i) {
          const unsigned long long cur = lkey[h];
          if (cur == k) {
            slot = (int)h;
            break;
          }
          if (cur == SDB_HKEY_EMPTY) {
            const unsigned long long prev =
              atomicCAS(&lkey[h], SDB_HKEY_EMPTY, k);
            if (prev == SDB_HKEY_EMPTY || prev == k) {
              slot = (int)h;
              break;
            }
          }
          h = (h + 1) & S_mask;
        }
      }
      if (slot >= 0) {
        hash_acc_add(a, &lacc[(uint32_t)slot * a.naggs], 0, rg, r0, r,
                     false);
        for (uint32_t q = 1; q < a.naggs; ++q)
          hash_acc_add(a, &lacc[(uint32_t)slot * a.naggs], q, rg, r0, r,
                       false);
      } else {
        const int64_t g = hash_global_upsert(a, k);
        if (g < 0) return;  // table overflow: host reports SDB_ERR_OOM
        for (uint32_t q = 0; q < a.naggs; ++q)
          hash_acc_add(a, &a.gacc[(uint64_t)g * a.naggs], q, rg, r0, r,
                       true);
      }
    }
  }

  unsigned long long wp = my_passed;
#pragma unroll
  for (int off = 32; off; off >>= 1) wp += __shfl_down(wp, off, 64);
  if ((threadIdx.x & 63) == 0 && wp) atomicAdd(a.rows_passed, wp);
  __syncthreads();
  // flush the LDS table into the global table
  for (uint32_t i = threadIdx.x; i < a.lds_slots; i += SCAN_NTHREADS) {
    const unsigned long long k = lkey[i];
    if (k == SDB_HKEY_EMPTY) continue;
    const int64_t g = hash_global_upsert(a, k);
    if (g < 0) return;
    for (uint32_t q = 0; q < a.naggs; ++q) {
      const unsigned long long v = lacc[i * a.naggs + q];
      if (a.agg_op[q] == SDB_AGG_SUM_F64) {
        double d;
        __builtin_memcpy(&d, &v, 8);
        if (d != 0.0)
          atomicAdd((double*)&a.gacc[(uint64_t)g * a.naggs + q], d);
      } else if (v) {
        atomicAdd(&a.gacc[(uint64_t)g * a.naggs + q], v);
      }
    }
  }
}

extern "C" {

static void table_free_str(SdbGpuTable* tab) {
  for (uint32_t k = 0; k < SDB_MAX_STRCOLS; ++k) {
    if (tab->str_off[k]) (void)hipFree(tab->str_off[k]);
    if (tab->str_blob[k]) (void)hipFree(tab->str_blob[k]);
    if (tab->str_mask[k]) (void)hipFree(tab->str_mask[k]);
  }
}

static void table_free_partial(SdbGpuTable* tab) {
  for (uint32_t c = 0; c < 16; ++c)
    if (tab->cols[c]) (void)hipFree(tab->cols[c]);
  table_free_str(tab);
  delete tab;
}

int sdb_gpu_table_load(SdbGpuCtx* ctx, const SdbColumnView* cols,
                       uint32_t ncols, uint64_t rows, SdbGpuTable** out) {
  if (!ctx || !cols || !out || ncols == 0 || ncols > 16)
    return SDB_ERR_INVALID;
  auto* tab = new SdbGpuTable{};
  tab->ncols = ncols;
  tab->rows = rows;
  tab->group_rows = 0;
  for (uint32_t c = 0; c < ncols; ++c) {
    tab->types[c] = cols[c].type;
    if (cols[c].type == SDB_COL_I64_FOR) {
      // encoded blob: upload whole blob, point refs into it
      SdbColHeaderDev hdr;
      std::memcpy(&hdr, cols[c].data, sizeof(hdr));
      // internal consistency (the view carries no external size, so the
      // declared extents must at least agree with themselves and `rows`)
      if (hdr.magic != SDB_COL_MAGIC_DEV || hdr.rows != rows ||
          hdr.group_rows == 0 ||
          hdr.ngroups != (rows + hdr.group_rows - 1) / hdr.group_rows ||
          hdr.off_desc > hdr.size || hdr.off_payload > hdr.size ||
          (uint64_t)hdr.ngroups * sizeof(SdbColGroupDescDev) >
            hdr.size - hdr.off_desc) {
        table_free_partial(tab);
        return SDB_ERR_INVALID;
      }
      if (tab->group_rows && tab->group_rows != hdr.group_rows) {
        table_free_partial(tab);  // FoR columns must share the group tiling
        return SDB_ERR_INVALID;
      }
      tab->group_rows = hdr.group_rows;
      // +4 pad bytes: bits_at reads one slack word past the final value
#define TAB_CHECK(x)                                         \
  do {                                                       \
    hipError_t _e = (x);                                     \
    if (_e != hipSuccess) {                                  \
      table_free_partial(tab);                               \
      return _e == hipErrorNoDevice ? SDB_ERR_NO_GPU         \
             : _e == hipErrorOutOfMemory ? SDB_ERR_OOM       \
                                         : SDB_ERR_HIP;      \
    }                                                        \
  } while (0)
      TAB_CHECK(hipMalloc(&tab->cols[c], hdr.size + 4));
      TAB_CHECK(hipMemcpy(tab->cols[c], cols[c].data, hdr.size,
                          hipMemcpyHostToDevice));
      tab->refs[c].data = (const uint8_t*)tab->cols[c] + hdr.off_payload;
      tab->refs[c].desc = (const SdbColGroupDescDev*)((const uint8_t*)
                            tab->cols[c] + hdr.off_desc);
      // host-side pass over the group table: max width sizes the staged
      // kernel's LDS budget; payword count bounds its slack loads
      const SdbColGroupDescDev* hdesc =
        (const SdbColGroupDescDev*)((const uint8_t*)cols[c].data +
                                    hdr.off_desc);
      uint16_t mw = 0;
      int64_t cmin = INT64_MAX, cmax = INT64_MIN;
      const uint64_t payw = (hdr.size - hdr.off_payload) / 4;
      for (uint32_t g = 0; g < hdr.ngroups; ++g) {
        const SdbColGroupDescDev& gd = hdesc[g];
        if (gd.vmin < cmin) cmin = gd.vmin;
        if (gd.vmax > cmax) cmax = gd.vmax;
        const uint64_t glen =
          std::min<uint64_t>(rows, (uint64_t)(g + 1) * hdr.group_rows) -
          (uint64_t)g * hdr.group_rows;
        if (gd.width > 32 ||
            (gd.width && (gd.word_off > payw ||
                          (glen * gd.width + 31) / 32 > payw - gd.word_off))) {
          table_free_partial(tab);
          return SDB_ERR_INVALID;
        }
        if (gd.width > mw) mw = gd.width;
      }
      tab->max_w[c] = mw;
      tab->paywords[c] = payw;
      tab->col_min[c] = cmin;
      tab->col_max[c] = cmax;
      tab->has_minmax[c] = hdr.ngroups > 0;
    } else {
      const size_t esz = cols[c].type == SDB_COL_I64 ? 8 : 4;
      TAB_CHECK(hipMalloc(&tab->cols[c], esz * rows));
      TAB_CHECK(hipMemcpy(tab->cols[c], cols[c].data, esz * rows,
                          hipMemcpyHostToDevice));
      tab->refs[c].data = tab->cols[c];
      tab->refs[c].desc = nullptr;
      if (cols[c].type == SDB_COL_I64 && rows) {
        // one ~HBM-rate pass at load; makes the raw-key range provable
        unsigned long long* d_mm;
        TAB_CHECK(hipMalloc(&d_mm, 16));
        const unsigned long long init[2] = {~0ull, 0ull};
        TAB_CHECK(hipMemcpy(d_mm, init, 16, hipMemcpyHostToDevice));
        const uint32_t nb =
          (uint32_t)std::min<uint64_t>(2048, (rows + 255) / 256);
        hipLaunchKernelGGL(col_minmax_kernel, dim3(nb), dim3(256), 0, 0,
                           (const long long*)tab->cols[c], rows, d_mm);
        unsigned long long h_mm[2];
        hipError_t e2 = hipMemcpy(h_mm, d_mm, 16, hipMemcpyDeviceToHost);
        (void)hipFree(d_mm);
        TAB_CHECK(e2);
        tab->col_min[c] = (int64_t)(h_mm[0] ^ 0x8000000000000000ull);
        tab->col_max[c] = (int64_t)(h_mm[1] ^ 0x8000000000000000ull);
        tab->has_minmax[c] = 1;
      }
    }
  }
#undef TAB_CHECK
  if (tab->group_rows == 0) tab->group_rows = 65536;  // raw-only tiling
  *out = tab;
  return SDB_OK;
}

int sdb_gpu_table_free(SdbGpuCtx* ctx, SdbGpuTable* tab) {
  if (!ctx || !tab) return SDB_ERR_INVALID;
  for (uint32_t c = 0; c < tab->ncols; ++c) {
    (void)hipFree(tab->cols[c]);
    if (tab->valid[c]) (void)hipFree(tab->valid[c]);
  }
  table_free_str(tab);
  delete tab;
  return SDB_OK;
}

int sdb_gpu_table_attach_validity(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                  uint32_t col, const uint64_t* bits) {
  if (!ctx || !tab || col >= tab->ncols) return SDB_ERR_INVALID;
  if (!bits) {
    if (tab->valid[col]) (void)hipFree(tab->valid[col]);
    tab->valid[col] = nullptr;
    return SDB_OK;
  }
  const uint64_t nwords = (tab->rows + 63) / 64;
  if (!tab->valid[col])
    HIP_CHECK(hipMalloc(&tab->valid[col], 8 * (nwords + 1)));
  HIP_CHECK(hipMemcpy(tab->valid[col], bits, 8 * nwords,
                      hipMemcpyHostToDevice));
  // synchronous for the same null-stream-ordering reason as the live
  // mask's pad word
  HIP_CHECK(hipMemset(tab->valid[col] + nwords, 0xFF, 8));
  return SDB_OK;
}

// ---- raw variable-width string columns (VERDICT r1 missing #5; the
// reference stores raw strings through its DuckDB fork's var-width
// vectors, column_reader.hpp:247-255 — parity at result level like the
// rest of the columnstore). Comparisons are memcmp order on unsigned
// bytes; predicates evaluate once into a row bitmask that scans consume
// through the existing validity-plane machinery (SDB_PRED_STRMASK). ----

int sdb_gpu_table_attach_strcol(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                uint32_t slot, const uint64_t* offsets,
                                const uint8_t* blob, uint64_t blob_len) {
  if (!ctx || !tab || slot >= SDB_MAX_STRCOLS || !offsets ||
      (!blob && blob_len))
    return SDB_ERR_INVALID;
  // offsets must be monotone and span exactly the blob
  if (offsets[tab->rows] != blob_len) return SDB_ERR_INVALID;
  for (uint64_t r = 0; r < tab->rows; ++r)
    if (offsets[r] > offsets[r + 1]) return SDB_ERR_INVALID;
  if (tab->str_off[slot]) (void)hipFree(tab->str_off[slot]);
  if (tab->str_blob[slot]) (void)hipFree(tab->str_blob[slot]);
  if (tab->str_mask[slot]) (void)hipFree(tab->str_mask[slot]);
  tab->str_off[slot] = nullptr;
  tab->str_blob[slot] = nullptr;
  tab->str_mask[slot] = nullptr;
  tab->str_mask_set[slot] = 0;
  tab->str_fsst[slot] = 0;
  HIP_CHECK(hipMalloc(&tab->str_off[slot], 8 * (tab->rows + 1)));
  HIP_CHECK(hipMemcpy(tab->str_off[slot], offsets, 8 * (tab->rows + 1),
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMalloc(&tab->str_blob[slot], blob_len ? blob_len : 1));
  if (blob_len)
    HIP_CHECK(hipMemcpy(tab->str_blob[slot], blob, blob_len,
                        hipMemcpyHostToDevice));
  tab->str_blob_len[slot] = blob_len;
  const uint64_t nwords = (tab->rows + 63) / 64;
  HIP_CHECK(hipMalloc(&tab->str_mask[slot], 8 * (nwords + 1)));
  return SDB_OK;
}

struct StrPredArgs {
  const uint64_t* off;
  const uint8_t* blob;
  uint64_t rows;
  unsigned long long* mask;
  uint32_t op;
  uint32_t lo_len, hi_len;
  uint8_t lo[SDB_STRLIT_MAX + 1], hi[SDB_STRLIT_MAX + 1];
};

// lexicographic compare, memcmp order on unsigned bytes
__device__ __forceinline__ int str_cmp_dev(const uint8_t* s, uint32_t n,
                                           const uint8_t* t, uint32_t m) {
  const uint32_t k = n < m ? n : m;
  for (uint32_t i = 0; i < k; ++i)
    if (s[i] != t[i]) return s[i] < t[i] ? -1 : 1;
  return n < m ? -1 : (n > m ? 1 : 0);
}

__global__ void strpred_kernel(StrPredArgs a) {
  // 64 consecutive rows per wave so one __ballot builds each mask word
  const uint64_t rpad = (a.rows + 63) & ~63ull;
  const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       r < rpad; r += stride) {
    bool m = false;
    if (r < a.rows) {
      const uint64_t o0 = a.off[r];
      const uint32_t len = (uint32_t)(a.off[r + 1] - o0);
      const uint8_t* sp = a.blob + o0;
      if (a.op == SDB_PRED_PREFIX) {
        m = len >= a.lo_len;
        for (uint32_t i = 0; m && i < a.lo_len; ++i) m = sp[i] == a.lo[i];
      } else {
        const int c = str_cmp_dev(sp, len, a.lo, a.lo_len);
        switch (a.op) {
          case SDB_PRED_LT: m = c < 0; break;
          case SDB_PRED_GE: m = c >= 0; break;
          case SDB_PRED_EQ: m = c == 0; break;
          case SDB_PRED_BETWEEN:
            m = c >= 0 && str_cmp_dev(sp, len, a.hi, a.hi_len) <= 0;
            break;
          default: break;
        }
      }
    }
    const unsigned long long w = __ballot(m);
    if ((threadIdx.x & 63u) == 0) a.mask[r >> 6] = w;
  }
}

struct StrPredFsstArgs {
  StrPredArgs base;
  uint32_t nsym;
  uint16_t symoff[256];
  uint8_t syms[2048];
};

// streaming comparator over the decoded byte stream: res 0 = equal so
// far, +-1 = decided by a byte inside the literal, 2 = row extends past
// an equal literal (row > literal). Final cmp derives at row end.
struct FsstCmp {
  int res;
  uint32_t pos;
};
__device__ __forceinline__ void fsst_feed(FsstCmp& st, uint8_t b,
                                          const uint8_t* lit,
                                          uint32_t litlen) {
  if (st.res) return;
  if (st.pos < litlen) {
    if (b != lit[st.pos]) st.res = b < lit[st.pos] ? -1 : 1;
    ++st.pos;
  } else {
    st.res = 2;
  }
}
__device__ __forceinline__ int fsst_final(const FsstCmp& st,
                                          uint32_t litlen) {
  if (st.res == 2) return 1;
  if (st.res) return st.res;
  return st.pos == litlen ? 0 : -1;  // proper prefix of literal => less
}

__global__ void strpred_fsst_kernel(StrPredFsstArgs fa) {
  __shared__ uint8_t ssym[2048];
  __shared__ uint16_t soff[256];
  for (uint32_t i = threadIdx.x; i < 2048; i += blockDim.x)
    ssym[i] = fa.syms[i];
  for (uint32_t i = threadIdx.x; i < 256; i += blockDim.x)
    soff[i] = fa.symoff[i];
  __syncthreads();
  const StrPredArgs a = fa.base;
  const uint64_t rpad = (a.rows + 63) & ~63ull;
  const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       r < rpad; r += stride) {
    bool m = false;
    if (r < a.rows) {
      const uint64_t o0 = a.off[r], o1 = a.off[r + 1];
      FsstCmp clo{0, 0}, chi{0, 0};
      bool malformed = false;
      for (uint64_t i = o0; i < o1; ++i) {
        const uint8_t c = a.blob[i];
        if (c == 255u) {  // escape: next byte is a literal
          if (++i >= o1) { malformed = true; break; }
          fsst_feed(clo, a.blob[i], a.lo, a.lo_len);
          if (a.op == SDB_PRED_BETWEEN)
            fsst_feed(chi, a.blob[i], a.hi, a.hi_len);
        } else {
          if (c >= fa.nsym) { malformed = true; break; }
          for (uint32_t j = soff[c]; j < (uint32_t)soff[c + 1]; ++j) {
            fsst_feed(clo, ssym[j], a.lo, a.lo_len);
            if (a.op == SDB_PRED_BETWEEN)
              fsst_feed(chi, ssym[j], a.hi, a.hi_len);
          }
        }
      }
      if (!malformed) {
        const int cl = fsst_final(clo, a.lo_len);
        switch (a.op) {
          case SDB_PRED_LT: m = cl < 0; break;
          case SDB_PRED_GE: m = cl >= 0; break;
          case SDB_PRED_EQ: m = cl == 0; break;
          case SDB_PRED_BETWEEN:
            m = cl >= 0 && fsst_final(chi, a.hi_len) <= 0;
            break;
          case SDB_PRED_PREFIX:
            // prefix iff no mismatch INSIDE the literal and the decoded
            // row reached the literal's end
            m = clo.res == 2 || (clo.res == 0 && clo.pos == a.lo_len);
            break;
          default: break;
        }
      }
    }
    const unsigned long long w = __ballot(m);
    if ((threadIdx.x & 63u) == 0) a.mask[r >> 6] = w;
  }
}

int sdb_gpu_strpred_mask(SdbGpuCtx* ctx, SdbGpuTable* tab, uint32_t slot,
                         SdbPredOp op, const uint8_t* lo, uint32_t lo_len,
                         const uint8_t* hi, uint32_t hi_len) {
  if (!ctx || !tab || slot >= SDB_MAX_STRCOLS || !tab->str_off[slot])
    return SDB_ERR_INVALID;
  if (op != SDB_PRED_LT && op != SDB_PRED_GE && op != SDB_PRED_BETWEEN &&
      op != SDB_PRED_EQ && op != SDB_PRED_PREFIX)
    return SDB_ERR_INVALID;
  if (lo_len > SDB_STRLIT_MAX || hi_len > SDB_STRLIT_MAX)
    return SDB_ERR_INVALID;
  if ((lo_len && !lo) || (hi_len && !hi)) return SDB_ERR_INVALID;
  if (op == SDB_PRED_BETWEEN && !hi && hi_len) return SDB_ERR_INVALID;
  StrPredArgs a{};
  a.off = tab->str_off[slot];
  a.blob = tab->str_blob[slot];
  a.rows = tab->rows;
  a.mask = tab->str_mask[slot];
  a.op = (uint32_t)op;
  a.lo_len = lo_len;
  a.hi_len = hi_len;
  if (lo_len) std::memcpy(a.lo, lo, lo_len);
  if (hi_len) std::memcpy(a.hi, hi, hi_len);
  const uint32_t nb =
    (uint32_t)std::min<uint64_t>(4096, (tab->rows + 255) / 256);
  if (tab->str_fsst[slot]) {
    StrPredFsstArgs fa{};
    std::memcpy(&fa.base, &a, sizeof(a));
    fa.nsym = tab->str_nsym[slot];
    std::memcpy(fa.symoff, tab->str_symoff[slot], sizeof(fa.symoff));
    std::memcpy(fa.syms, tab->str_syms[slot], sizeof(fa.syms));
    hipLaunchKernelGGL(strpred_fsst_kernel, dim3(nb ? nb : 1), dim3(256),
                       0, ctx->stream, fa);
  } else {
    hipLaunchKernelGGL(strpred_kernel, dim3(nb ? nb : 1), dim3(256), 0,
                       ctx->stream, a);
  }
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  tab->str_mask_set[slot] = 1;
  return SDB_OK;
}

// FSST-style compressed string slot (second half of
// "dict/FSST-style"; the reference defers string codecs to its DuckDB
// fork's COMPRESSION_AUTO, column_writer.cpp:141-204 — parity at result
// level). Encoded rows expand deterministically, so predicates decode on
// the fly and never materialize the string.
int sdb_gpu_table_attach_strcol_fsst(SdbGpuCtx* ctx, SdbGpuTable* tab,
                                     uint32_t slot,
                                     const uint64_t* offsets,
                                     const uint8_t* enc_blob,
                                     uint64_t enc_len,
                                     const uint8_t* symbols,
                                     const uint32_t* sym_offsets,
                                     uint32_t nsym) {
  if (nsym > 254 || !sym_offsets || (!symbols && nsym))
    return SDB_ERR_INVALID;
  if (sym_offsets[nsym] > 2048) return SDB_ERR_INVALID;
  for (uint32_t i = 0; i < nsym; ++i) {
    const uint32_t l = sym_offsets[i + 1] - sym_offsets[i];
    if (sym_offsets[i] > sym_offsets[i + 1] || l == 0 || l > 8)
      return SDB_ERR_INVALID;
  }
  // escape codes must reference in-bounds literals: enforced by the
  // offsets-monotone check (an escape at a row end would make the row's
  // decode read past its span; validated at predicate time per row)
  const int rc = sdb_gpu_table_attach_strcol(ctx, tab, slot, offsets,
                                             enc_blob, enc_len);
  if (rc) return rc;
  tab->str_fsst[slot] = 1;
  tab->str_nsym[slot] = (uint16_t)nsym;
  std::memset(tab->str_symoff[slot], 0, sizeof(tab->str_symoff[slot]));
  std::memset(tab->str_syms[slot], 0, sizeof(tab->str_syms[slot]));
  for (uint32_t i = 0; i <= nsym; ++i)
    tab->str_symoff[slot][i] = (uint16_t)sym_offsets[i];
  if (nsym)
    std::memcpy(tab->str_syms[slot], symbols, sym_offsets[nsym]);
  return SDB_OK;
}

int sdb_gpu_scan_agg(SdbGpuCtx* ctx, SdbGpuTable* tab, uint32_t group_col,
                     uint32_t ngroups, const SdbPredSpec* preds,
                     uint32_t npreds, const SdbAggSpec* aggs, uint32_t naggs,
                     SdbAggResult* out, uint64_t* rows_passed) {
  if (!ctx || !tab || !out || !rows_passed || group_col >= tab->ncols ||
      ngroups == 0 || ngroups > SCAN_MAX_GROUPS || naggs == 0 ||
      naggs > SCAN_MAX_AGGS || npreds > SCAN_MAX_PREDS)
    return SDB_ERR_INVALID;
  if (tab->types[group_col] == SDB_COL_F32) return SDB_ERR_INVALID;
  if (tab->valid[group_col]) return SDB_ERR_INVALID;  // no NULL groups
  // dense-key contract: keys must be provably inside [0, ngroups) or the
  // perfect-hash accumulate acc[key*naggs+q] corrupts LDS. Both column
  // kinds carry a load-time range (FoR zonemaps / raw-column reduction).
  if (tab->rows && tab->has_minmax[group_col] &&
      (tab->col_min[group_col] < 0 ||
       tab->col_max[group_col] >= (int64_t)ngroups))
    return SDB_ERR_INVALID;

  hipStream_t stream = ctx->stream;

  ScanArgs a{};
  a.keys = tab->refs[group_col];
  a.rows = tab->rows;
  a.group_rows = tab->group_rows;
  a.ngroups = ngroups;
  a.naggs = naggs;
  a.npreds = npreds;
  for (uint32_t p = 0; p < npreds; ++p) {
    if (preds[p].op == SDB_PRED_STRMASK) {
      // raw-string predicate: consume the slot's precomputed row bitmask
      // through the validity machinery — NOTNULL evaluates the plane
      // alone; the value column is the (already resident) group key
      const uint32_t slot = preds[p].col;
      if (slot >= SDB_MAX_STRCOLS || !tab->str_mask_set[slot])
        return SDB_ERR_INVALID;
      a.pred_col[p] = tab->refs[group_col];
      a.pred_op[p] = SDB_PRED_NOTNULL;
      a.pred_isf32[p] = 0;
      a.pred_valid[p] = tab->str_mask[slot];
      continue;
    }
    if (preds[p].col >= tab->ncols) return SDB_ERR_INVALID;
    if (preds[p].op < SDB_PRED_LT || preds[p].op > SDB_PRED_NOTNULL)
      return SDB_ERR_INVALID;
    a.pred_col[p] = tab->refs[preds[p].col];
    a.pred_op[p] = preds[p].op;
    a.pred_isf32[p] = tab->types[preds[p].col] == SDB_COL_F32 ? 1 : 0;
    a.pred_lo[p] = preds[p].ilo;
    a.pred_hi[p] = preds[p].ihi;
    a.pred_flo[p] = preds[p].flo;
    a.pred_fhi[p] = preds[p].fhi;
    a.pred_valid[p] = tab->valid[preds[p].col];
  }
  for (uint32_t q = 0; q < naggs; ++q) {
    a.agg_op[q] = aggs[q].op;
    a.agg_col[q] = aggs[q].op == SDB_AGG_COUNT ? ColRef{nullptr, nullptr}
                                               : tab->refs[aggs[q].col];
    a.agg_valid[q] =
      aggs[q].op == SDB_AGG_COUNT ? nullptr : tab->valid[aggs[q].col];
    a.agg_src[q] = 0;
    if (aggs[q].op == SDB_AGG_SUM_I64) {
      if (aggs[q].col == group_col) a.agg_src[q] = 9;
      for (uint32_t p = 0; p < npreds; ++p)
        if (preds[p].op != SDB_PRED_STRMASK &&  // str preds carry a SLOT
            aggs[q].col == preds[p].col)
          a.agg_src[q] = 1 + (int)p;
    }
    if (aggs[q].op == SDB_AGG_SUM_I64 &&
        tab->types[aggs[q].col] == SDB_COL_F32)
      return SDB_ERR_INVALID;
    if (aggs[q].op == SDB_AGG_SUM_F64 &&
        tab->types[aggs[q].col] != SDB_COL_F32)
      return SDB_ERR_INVALID;
  }
  const uint32_t nslots = ngroups * naggs;
  // ctx-cached workspace: the dense scan hot path allocates nothing per
  // call (grown on demand; buffers owned and freed by the context)
#define HIP_CHECK_CLEAN(x)                                   \
  do {                                                       \
    hipError_t _e = (x);                                     \
    if (_e != hipSuccess) {                                  \
      return _e == hipErrorNoDevice ? SDB_ERR_NO_GPU         \
             : _e == hipErrorOutOfMemory ? SDB_ERR_OOM       \
                                         : SDB_ERR_HIP;      \
    }                                                        \
  } while (0)
  if (ctx->scan_out_cap < nslots) {
    if (ctx->d_scan_out) (void)hipFree(ctx->d_scan_out);
    ctx->d_scan_out = nullptr;
    ctx->scan_out_cap = 0;
    HIP_CHECK_CLEAN(hipMalloc(&ctx->d_scan_out, 8ull * nslots));
    ctx->scan_out_cap = nslots;
  }
  if (!ctx->d_scan_passed)
    HIP_CHECK_CLEAN(hipMalloc(&ctx->d_scan_passed, 8));
  unsigned long long* d_out = ctx->d_scan_out;
  unsigned long long* d_passed = ctx->d_scan_passed;
  HIP_CHECK_CLEAN(hipMemsetAsync(d_out, 0, 8ull * nslots, stream));
  HIP_CHECK_CLEAN(hipMemsetAsync(d_passed, 0, 8, stream));
  a.out = d_out;
  a.rows_passed = d_passed;

  uint32_t nblocks =
    (uint32_t)((tab->rows + a.group_rows - 1) / a.group_rows);
  if (nblocks > SCAN_MAXB) nblocks = SCAN_MAXB;
  if (nblocks < 1) nblocks = 1;
  size_t lds = 8ull * nslots;
  bool any_for = a.keys.desc != nullptr;
  for (uint32_t p = 0; p < npreds; ++p) any_for |= a.pred_col[p].desc != nullptr;
  for (uint32_t q = 0; q < naggs; ++q) any_for |= a.agg_col[q].desc != nullptr;

  // staged-FoR eligibility: distinct FoR columns among {keys, pred0, pred1}
  // get LDS chunk staging; the chunk is shrunk until acc + stage buffers fit
  // the 160 KB LDS. SDB_SCAN_NOSTAGE falls back to the unstaged walker.
  uint32_t st_col[3];
  a.nstage = 0;
  a.key_st = -1;
  a.pred_st[0] = a.pred_st[1] = -1;
  auto add_stage = [&](uint32_t col) -> int {
    if (!tab->refs[col].desc) return -1;
    for (uint32_t s = 0; s < a.nstage; ++s)
      if (st_col[s] == col) return (int)s;
    st_col[a.nstage] = col;
    return (int)a.nstage++;
  };
  a.key_st = add_stage(group_col);
  for (uint32_t p = 0; p < npreds && p < 2; ++p)
    a.pred_st[p] = add_stage(preds[p].col);
  bool staged = any_for && a.nstage > 0 && !getenv("SDB_SCAN_NOSTAGE");
  if (staged) {
    uint32_t C = 8192;  // rows per chunk; multiple of 32 (word alignment)
    size_t ldsb = 0;
    for (;;) {
      ldsb = (8ull * nslots + 15) & ~15ull;
      for (uint32_t s = 0; s < a.nstage; ++s) {
        a.st_lds_off[s] = (uint32_t)(ldsb / 4);
        const uint64_t capw = (uint64_t)C * tab->max_w[st_col[s]] / 32 + 4;
        ldsb = (ldsb + 4 * capw + 15) & ~15ull;
      }
      if (ldsb <= 160 * 1024 || C <= 1024) break;
      C >>= 1;
    }
    if (ldsb > 160 * 1024) {
      staged = false;  // huge group table: unstaged walker
    } else {
      a.chunk_rows = C;
      for (uint32_t s = 0; s < a.nstage; ++s) {
        a.st_desc[s] = tab->refs[st_col[s]].desc;
        a.st_pay[s] = (const uint32_t*)tab->refs[st_col[s]].data;
        a.st_paywords[s] = tab->paywords[st_col[s]];
      }
      a.f32_pay = nullptr;
      if (!getenv("SDB_SCAN_NOF32STAGE") &&
          ldsb + 4ull * C + 16 <= 160 * 1024) {
        for (uint32_t q = 0; q < naggs; ++q) {
          if (a.agg_op[q] == SDB_AGG_SUM_F64 && !a.agg_col[q].desc) {
            a.f32_pay = (const float*)a.agg_col[q].data;
            a.f32_lds_off = (uint32_t)(ldsb / 4);
            ldsb = (ldsb + 4ull * C + 15) & ~15ull;
            break;
          }
        }
      }
      lds = ldsb;
    }
  }

  // compile-time agg-shape dispatch: the common [COUNT, SUM_I64,
  // SUM_F64] shape gets the straight-lined FoR accumulate bodies
  const bool shape3 = naggs == 3 && aggs[0].op == SDB_AGG_COUNT &&
                      aggs[1].op == SDB_AGG_SUM_I64 &&
                      aggs[2].op == SDB_AGG_SUM_F64 &&
                      !getenv("SDB_SCAN_NOSHAPE");
  if (staged && shape3)
    hipLaunchKernelGGL((scan_agg_staged_kernel<1>), dim3(nblocks),
                       dim3(SCAN_NTHREADS), lds, stream, a);
  else if (staged)
    hipLaunchKernelGGL((scan_agg_staged_kernel<0>), dim3(nblocks),
                       dim3(SCAN_NTHREADS), lds, stream, a);
  else if (any_for && shape3)
    hipLaunchKernelGGL((scan_agg_kernel<0, 1>), dim3(nblocks),
                       dim3(SCAN_NTHREADS), lds, stream, a);
  else if (any_for)
    hipLaunchKernelGGL((scan_agg_kernel<0, 0>), dim3(nblocks),
                       dim3(SCAN_NTHREADS), lds, stream, a);
  else
    // RAW stays generic: the straight-lined 3-agg arm measured 183G vs
    // 234G rows/s on the same box (r2_raw_shape_ab.log) — the runtime
    // agg loop bounds the e-unroll better there; only the
    // instruction-bound FoR walkers keep ASHAPE=1
    hipLaunchKernelGGL((scan_agg_kernel<1, 0>), dim3(nblocks),
                       dim3(SCAN_NTHREADS), lds, stream, a);
  HIP_CHECK_CLEAN(hipGetLastError());
  std::vector<unsigned long long> h_out(nslots);
  unsigned long long h_passed = 0;
  HIP_CHECK_CLEAN(hipMemcpyAsync(h_out.data(), d_out, 8ull * nslots,
                                 hipMemcpyDeviceToHost, stream));
  HIP_CHECK_CLEAN(hipMemcpyAsync(&h_passed, d_passed, 8,
                                 hipMemcpyDeviceToHost, stream));
  HIP_CHECK_CLEAN(hipStreamSynchronize(stream));
#undef HIP_CHECK_CLEAN
  // d_out/d_passed are ctx-cached; freed with the context
  for (uint32_t g = 0; g < ngroups; ++g) {
    for (uint32_t q = 0; q < naggs; ++q) {
      const unsigned long long raw = h_out[g * naggs + q];
      SdbAggResult* r = &out[g * naggs + q];
      if (aggs[q].op == SDB_AGG_SUM_F64) {
        double v;
        std::memcpy(&v, &raw, 8);
        r->f64 = v;
        r->i64 = 0;
      } else {
        r->i64 = (int64_t)raw;
        r->f64 = 0;
      }
    }
  }
  *rows_passed = h_passed;
  return SDB_OK;
}


// core shared by the i64-key entry and the string-key entry: gkeys is
// the group-key column (a table column or a derived device array)
static int scan_agg_hash_core(SdbGpuCtx* ctx, SdbGpuTable* tab,
                              ColRef gkeys, uint64_t max_groups,
                              const SdbPredSpec* preds, uint32_t npreds,
                              const SdbAggSpec* aggs, uint32_t naggs,
                              int64_t* keys_out, SdbAggResult* out,
                              uint64_t* ngroups_out,
                              uint64_t* rows_passed) {
  if (!ctx || !tab || !keys_out || !out || !ngroups_out || !rows_passed ||
      max_groups == 0 || max_groups > (1u << 22) || naggs == 0 ||
      naggs > SCAN_MAX_AGGS || npreds > SCAN_MAX_PREDS)
    return SDB_ERR_INVALID;

  hipStream_t stream = ctx->stream;
  HashAggArgs a{};
  a.keys = gkeys;
  a.rows = tab->rows;
  a.group_rows = tab->group_rows;
  a.naggs = naggs;
  a.npreds = npreds;
  for (uint32_t p = 0; p < npreds; ++p) {
    if (preds[p].op == SDB_PRED_STRMASK) {
      // raw-string predicate: consume the slot's precomputed row bitmask
      // through the validity machinery — NOTNULL evaluates the plane
      // alone; the value column is the (already resident) group key
      const uint32_t slot = preds[p].col;
      if (slot >= SDB_MAX_STRCOLS || !tab->str_mask_set[slot])
        return SDB_ERR_INVALID;
      a.pred_col[p] = gkeys;
      a.pred_op[p] = SDB_PRED_NOTNULL;
      a.pred_isf32[p] = 0;
      a.pred_valid[p] = tab->str_mask[slot];
      continue;
    }
    if (preds[p].col >= tab->ncols) return SDB_ERR_INVALID;
    if (preds[p].op < SDB_PRED_LT || preds[p].op > SDB_PRED_NOTNULL)
      return SDB_ERR_INVALID;
    a.pred_col[p] = tab->refs[preds[p].col];
    a.pred_op[p] = preds[p].op;
    a.pred_isf32[p] = tab->types[preds[p].col] == SDB_COL_F32 ? 1 : 0;
    a.pred_lo[p] = preds[p].ilo;
    a.pred_hi[p] = preds[p].ihi;
    a.pred_flo[p] = preds[p].flo;
    a.pred_fhi[p] = preds[p].fhi;
    a.pred_valid[p] = tab->valid[preds[p].col];
  }
  for (uint32_t q = 0; q < naggs; ++q) {
    a.agg_op[q] = aggs[q].op;
    a.agg_col[q] = aggs[q].op == SDB_AGG_COUNT ? ColRef{nullptr, nullptr}
                                               : tab->refs[aggs[q].col];
    a.agg_valid[q] =
      aggs[q].op == SDB_AGG_COUNT ? nullptr : tab->valid[aggs[q].col];
    if (aggs[q].op == SDB_AGG_SUM_I64 &&
        tab->types[aggs[q].col] == SDB_COL_F32)
      return SDB_ERR_INVALID;
    if (aggs[q].op == SDB_AGG_SUM_F64 &&
        tab->types[aggs[q].col] != SDB_COL_F32)
      return SDB_ERR_INVALID;
  }
  // global table: pow2 capacity >= 2*max_groups (load factor <= 0.5)
  uint64_t cap = 64;
  while (cap < 2 * max_groups) cap <<= 1;
  a.gcap_mask = (uint32_t)(cap - 1);
  a.max_groups = (uint32_t)max_groups;
  // LDS: largest pow2 slot count whose (key + naggs accumulators) fit.
  // When the distinct-key bound dwarfs the LDS table, nearly every row
  // would burn a full probe chain before falling through to the global
  // table anyway (measured 3.3G rows/s at 100k groups vs 67.9G dense):
  // skip LDS staging entirely and upsert straight into the global table.
  uint32_t slots = 1u << 14;
  while (slots && (uint64_t)slots * 8 * (1 + naggs) > 150 * 1024)
    slots >>= 1;
  if (max_groups > (uint64_t)slots * 2) slots = 0;
  a.lds_slots = slots;
  const size_t lds = slots ? (size_t)slots * 8 * (1 + naggs) : 16;

  unsigned long long* d_keys = nullptr;
  unsigned long long* d_acc = nullptr;
  unsigned long long* d_misc = nullptr;  // [0]=inserts|overflow, then
                                         // neg_acc[naggs+1], rows_passed
  const size_t misc_words = 1 + (naggs + 1) + 1;
#define HASH_CHECK(x)                                        \
  do {                                                       \
    hipError_t _e = (x);                                     \
    if (_e != hipSuccess) {                                  \
      if (d_keys) (void)hipFree(d_keys);                     \
      if (d_acc) (void)hipFree(d_acc);                       \
      if (d_misc) (void)hipFree(d_misc);                     \
      return _e == hipErrorNoDevice ? SDB_ERR_NO_GPU         \
             : _e == hipErrorOutOfMemory ? SDB_ERR_OOM       \
                                         : SDB_ERR_HIP;      \
    }                                                        \
  } while (0)
  HASH_CHECK(hipMalloc(&d_keys, cap * 8));
  HASH_CHECK(hipMalloc(&d_acc, cap * 8 * naggs));
  HASH_CHECK(hipMalloc(&d_misc, misc_words * 8));
  HASH_CHECK(hipMemsetAsync(d_keys, 0xFF, cap * 8, stream));
  HASH_CHECK(hipMemsetAsync(d_acc, 0, cap * 8 * naggs, stream));
  HASH_CHECK(hipMemsetAsync(d_misc, 0, misc_words * 8, stream));
  a.gkeys = d_keys;
  a.gacc = d_acc;
  a.ginserts = (uint32_t*)d_misc;
  a.goverflow = (uint32_t*)d_misc + 1;
  a.neg_acc = d_misc + 1;
  a.rows_passed = d_misc + 1 + (naggs + 1);

  uint32_t nblocks =
    (uint32_t)((tab->rows + a.group_rows - 1) / a.group_rows);
  if (nblocks > SCAN_MAXB) nblocks = SCAN_MAXB;
  if (nblocks < 1) nblocks = 1;
  hipLaunchKernelGGL(scan_agg_hash_kernel, dim3(nblocks),
                     dim3(SCAN_NTHREADS), lds, stream, a);
  HASH_CHECK(hipGetLastError());

  std::vector<unsigned long long> h_keys(cap);
  std::vector<unsigned long long> h_acc(cap * naggs);
  std::vector<unsigned long long> h_misc(misc_words);
  HASH_CHECK(hipMemcpyAsync(h_keys.data(), d_keys, cap * 8,
                            hipMemcpyDeviceToHost, stream));
  HASH_CHECK(hipMemcpyAsync(h_acc.data(), d_acc, cap * 8 * naggs,
                            hipMemcpyDeviceToHost, stream));
  HASH_CHECK(hipMemcpyAsync(h_misc.data(), d_misc, misc_words * 8,
                            hipMemcpyDeviceToHost, stream));
  HASH_CHECK(hipStreamSynchronize(stream));
#undef HASH_CHECK
  (void)hipFree(d_keys);
  (void)hipFree(d_acc);
  (void)hipFree(d_misc);

  const uint32_t inserts = (uint32_t)(h_misc[0] & 0xFFFFFFFFu);
  const uint32_t overflow = (uint32_t)(h_misc[0] >> 32);
  const bool neg_present = h_misc[1 + naggs] != 0;
  const uint64_t distinct = (uint64_t)inserts + (neg_present ? 1 : 0);
  if (overflow || distinct > max_groups) return SDB_ERR_OOM;

  // compact + sort by key ascending (deterministic result order)
  std::vector<uint32_t> live;
  live.reserve(inserts);
  for (uint64_t i = 0; i < cap; ++i)
    if (h_keys[i] != SDB_HKEY_EMPTY) live.push_back((uint32_t)i);
  std::sort(live.begin(), live.end(), [&](uint32_t x, uint32_t y) {
    return (int64_t)h_keys[x] < (int64_t)h_keys[y];
  });
  uint64_t n = 0;
  auto emit = [&](int64_t key, const unsigned long long* acc) {
    keys_out[n] = key;
    for (uint32_t q = 0; q < naggs; ++q) {
      SdbAggResult* r = &out[n * naggs + q];
      if (aggs[q].op == SDB_AGG_SUM_F64) {
        double v;
        std::memcpy(&v, &acc[q], 8);
        r->f64 = v;
        r->i64 = 0;
      } else {
        r->i64 = (int64_t)acc[q];
        r->f64 = 0;
      }
    }
    ++n;
  };
  // negative keys sort before -1? No: -1 is the LARGEST negative value,
  // so emit live keys < -1 first, then -1, then the rest.
  size_t li = 0;
  while (li < live.size() && (int64_t)h_keys[live[li]] < -1)
    emit((int64_t)h_keys[live[li]], &h_acc[(uint64_t)live[li] * naggs]),
      ++li;
  if (neg_present) emit(-1, &h_misc[1]);
  for (; li < live.size(); ++li)
    emit((int64_t)h_keys[live[li]], &h_acc[(uint64_t)live[li] * naggs]);
  *ngroups_out = n;
  *rows_passed = h_misc[1 + naggs + 1];
  return SDB_OK;
}

int sdb_gpu_scan_agg_hash(SdbGpuCtx* ctx, SdbGpuTable* tab,
                          uint32_t group_col, uint64_t max_groups,
                          const SdbPredSpec* preds, uint32_t npreds,
                          const SdbAggSpec* aggs, uint32_t naggs,
                          int64_t* keys_out, SdbAggResult* out,
                          uint64_t* ngroups_out, uint64_t* rows_passed) {
  if (!ctx || !tab || group_col >= tab->ncols) return SDB_ERR_INVALID;
  if (tab->types[group_col] == SDB_COL_F32) return SDB_ERR_INVALID;
  if (tab->valid[group_col]) return SDB_ERR_INVALID;  // no NULL groups
  return scan_agg_hash_core(ctx, tab, tab->refs[group_col], max_groups,
                            preds, npreds, aggs, naggs, keys_out, out,
                            ngroups_out, rows_passed);
}

// FNV-1a 64 over each row's DECODED string bytes (raw slot: the bytes
// themselves; FSST slot: symbols expanded on the fly). Deterministic and
// restated in oracle/pyoracle.fnv1a64 — the string-key GROUP BY below
// groups by this hash, with injectivity over the column's distinct
// strings checked host-side by the caller (collision => exact fallback).
struct StrHashArgs {
  const uint64_t* off;
  const uint8_t* blob;
  uint64_t rows;
  long long* out;
  uint32_t nsym;  // 0 = raw slot
  uint16_t symoff[256];
  uint8_t syms[2048];
};

__global__ void strhash_kernel(StrHashArgs ha) {
  __shared__ uint8_t ssym[2048];
  __shared__ uint16_t soff[256];
  for (uint32_t i = threadIdx.x; i < 2048; i += blockDim.x)
    ssym[i] = ha.syms[i];
  for (uint32_t i = threadIdx.x; i < 256; i += blockDim.x)
    soff[i] = ha.symoff[i];
  __syncthreads();
  const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       r < ha.rows; r += stride) {
    const uint64_t o0 = ha.off[r], o1 = ha.off[r + 1];
    unsigned long long h = 14695981039346656037ull;
    bool bad = false;
    if (ha.nsym == 0) {  // raw bytes
      for (uint64_t i = o0; i < o1; ++i) {
        h ^= ha.blob[i];
        h *= 1099511628211ull;
      }
    } else {
      for (uint64_t i = o0; i < o1; ++i) {
        const uint8_t c = ha.blob[i];
        if (c == 255u) {
          if (++i >= o1) { bad = true; break; }
          h ^= ha.blob[i];
          h *= 1099511628211ull;
        } else if (c >= ha.nsym) {
          bad = true;
          break;
        } else {
          for (uint32_t j = soff[c]; j < (uint32_t)soff[c + 1]; ++j) {
            h ^= ssym[j];
            h *= 1099511628211ull;
          }
        }
      }
    }
    // malformed rows hash to the FNV basis of the empty string XOR a
    // sentinel so they cannot silently merge with a real group
    ha.out[r] = bad ? (long long)0x8000000000000001ull : (long long)h;
  }
}

// GROUP BY string keys ("GROUP BY text keys" on
// non-dictionary columns; the reference hands this to its DuckDB fork's
// PhysicalHashAggregate over string vectors — result-level parity).
// keys_out receives the FNV-1a 64 hash of each group's string; the
// caller resolves hashes back to strings host-side and MUST verify
// injectivity over the column's distinct strings (the python wrapper
// does both; a collision is detected, never silent).
int sdb_gpu_scan_agg_hash_str(SdbGpuCtx* ctx, SdbGpuTable* tab,
                              uint32_t str_slot, uint64_t max_groups,
                              const SdbPredSpec* preds, uint32_t npreds,
                              const SdbAggSpec* aggs, uint32_t naggs,
                              int64_t* keys_out, SdbAggResult* out,
                              uint64_t* ngroups_out,
                              uint64_t* rows_passed) {
  if (!ctx || !tab || str_slot >= SDB_MAX_STRCOLS ||
      !tab->str_off[str_slot])
    return SDB_ERR_INVALID;
  long long* d_hash = nullptr;
  HIP_CHECK(hipMalloc(&d_hash, 8ull * (tab->rows ? tab->rows : 1)));
  StrHashArgs ha{};
  ha.off = tab->str_off[str_slot];
  ha.blob = tab->str_blob[str_slot];
  ha.rows = tab->rows;
  ha.out = d_hash;
  if (tab->str_fsst[str_slot]) {
    ha.nsym = tab->str_nsym[str_slot];
    std::memcpy(ha.symoff, tab->str_symoff[str_slot], sizeof(ha.symoff));
    std::memcpy(ha.syms, tab->str_syms[str_slot], sizeof(ha.syms));
  }
  const uint32_t nb =
    (uint32_t)std::min<uint64_t>(4096, (tab->rows + 255) / 256);
  hipLaunchKernelGGL(strhash_kernel, dim3(nb ? nb : 1), dim3(256), 0,
                     ctx->stream, ha);
  hipError_t le = hipGetLastError();
  if (le != hipSuccess) {
    (void)hipFree(d_hash);
    return SDB_ERR_HIP;
  }
  ColRef gk{d_hash, nullptr};
  const int rc = scan_agg_hash_core(ctx, tab, gk, max_groups, preds,
                                    npreds, aggs, naggs, keys_out, out,
                                    ngroups_out, rows_passed);
  (void)hipFree(d_hash);
  return rc;
}


}  // extern "C"
