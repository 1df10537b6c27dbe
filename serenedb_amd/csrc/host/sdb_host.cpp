// sdb_host.cpp — host-side (index-build + plan-prep) part of the
// MI355X-native SereneDB hot path.
//
// Contents:
//   * FormatTraits128-compatible block codec (encode + scalar decode):
//     restates libs/iresearch/include/iresearch/formats/posting/
//     format_block_128.hpp:51-379 (write) and :446-636 (read); delta-bitpack
//     bit layout per third_party/simdcomp simdpackwithoutmaskd1/simdunpackd1
//     (see include/sdb_format.h header comment for the layout spec).
//   * Synthetic segment builder (the index-write side is OUT OF SCOPE per
//     SURVEY.md §2 — CPU-built synthetic segments feed the GPU; this builder
//     is the project's replacement for the reference's index_writer at test/
//     bench time).
//   * Seeded synthetic corpus generator (SURVEY.md §8d distributions).
//   * BM25 stats preparation (double->f32 exactly as bm25.cpp:288-306).
//   * Final host-side top-k select (the PrepareEmitBuffer analogue,
//     server/connector/duckdb_search_full_scan.cpp:1945-2000).
//
// This library is pure host C++ (no HIP): it must run both in the CPU-only
// build container and on the GPU box. The QUERY path lives in libsdb_gpu
// (csrc/host/api.cpp + csrc/hip/*) and fails loudly without a GPU.
//
// NOTE: compiled with -ffp-contract=off so fp32 BM25 arithmetic is
// bit-identical across gcc/hipcc/oracle (DESIGN.md "Determinism").

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "../../../include/sdb_format.h"

namespace {

constexpr uint32_t kBlock = SDB_BLOCK_SIZE;

// ---------------------------------------------------------------------------
// bit utils
// ---------------------------------------------------------------------------
inline uint32_t bit_width_u32(uint32_t v) {
  return v == 0 ? 0 : 32u - __builtin_clz(v);
}

// ByteSize1234: format_block_128.hpp:784-795
inline uint32_t byte_size_1234(uint32_t v) {
  if (v < (1u << 8)) return 1;
  if (v < (1u << 16)) return 2;
  if (v < (1u << 24)) return 3;
  return 4;
}
// ByteSize0124: format_block_128.hpp:797-808
inline uint32_t byte_size_0124(uint32_t v) {
  if (v == 0) return 0;
  if (v < (1u << 8)) return 1;
  if (v < (1u << 16)) return 2;
  return 4;
}

// ---------------------------------------------------------------------------
// simdcomp-layout vertical bitpack (scalar restatement)
// value index i -> SSE lane c = i&3, group g = i>>2; lane c's groups form an
// LSB-first bitstream packed into 32-bit words; flat word w*4+c is word w of
// lane c. (simdintegratedbitpacking.c ipackwithoutmaskN / iunpackN;
// simdbitpacking.c __SIMD_fastpackwithoutmask_N / __SIMD_fastunpack_N)
// ---------------------------------------------------------------------------
static void pack_vertical(const uint32_t* vals /*128*/, uint32_t bits,
                          uint8_t* out /*16*bits bytes*/) {
  std::memset(out, 0, 16u * bits);
  auto* w = reinterpret_cast<uint32_t*>(out);
  for (uint32_t i = 0; i < kBlock; ++i) {
    const uint32_t c = i & 3u, g = i >> 2;
    const uint64_t bitpos = uint64_t(g) * bits;
    const uint32_t word = uint32_t(bitpos >> 5), sh = uint32_t(bitpos & 31);
    const uint64_t v = uint64_t(vals[i]) << sh;
    w[word * 4 + c] |= uint32_t(v);
    if (sh + bits > 32) w[(word + 1) * 4 + c] |= uint32_t(v >> 32);
  }
}

static void unpack_vertical(const uint8_t* in, uint32_t bits,
                            uint32_t* vals /*128*/) {
  const uint32_t mask = bits == 32 ? 0xFFFFFFFFu : ((1u << bits) - 1u);
  const uint32_t* w = reinterpret_cast<const uint32_t*>(in);
  // in may be unaligned in the payload stream: memcpy-safe word reads
  auto rdw = [&](uint32_t idx) {
    uint32_t x;
    std::memcpy(&x, reinterpret_cast<const uint8_t*>(w) + 4u * idx, 4);
    return x;
  };
  for (uint32_t i = 0; i < kBlock; ++i) {
    const uint32_t c = i & 3u, g = i >> 2;
    const uint64_t bitpos = uint64_t(g) * bits;
    const uint32_t word = uint32_t(bitpos >> 5), sh = uint32_t(bitpos & 31);
    uint64_t v = rdw(word * 4 + c) >> sh;
    if (sh + bits > 32) v |= uint64_t(rdw((word + 1) * 4 + c)) << (32 - sh);
    vals[i] = uint32_t(v) & mask;
  }
}

// ---------------------------------------------------------------------------
// streamvbyte 1234 (public Lemire format; format_block_128.hpp:182-199,
// 507-517 call sites). Control bytes first (2-bit codes, low bits = first
// value, code = nbytes-1), then little-endian data bytes.
// ---------------------------------------------------------------------------
static uint32_t svb_encode(const uint32_t* vals, uint32_t len, uint8_t* out) {
  const uint32_t groups = (len + 3) / 4;
  uint8_t* ctrl = out;
  uint8_t* data = out + groups;
  std::memset(ctrl, 0, groups);
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t n = byte_size_1234(vals[i]);
    ctrl[i >> 2] |= uint8_t((n - 1) << ((i & 3) * 2));
    uint32_t v = vals[i];
    for (uint32_t b = 0; b < n; ++b) {
      *data++ = uint8_t(v);
      v >>= 8;
    }
  }
  return uint32_t(data - out);
}

static uint32_t svb_decode(const uint8_t* in, uint32_t* vals, uint32_t len) {
  const uint32_t groups = (len + 3) / 4;
  const uint8_t* ctrl = in;
  const uint8_t* data = in + groups;
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t n = ((ctrl[i >> 2] >> ((i & 3) * 2)) & 3u) + 1u;
    uint32_t v = 0;
    for (uint32_t b = 0; b < n; ++b) v |= uint32_t(*data++) << (8 * b);
    vals[i] = v;
  }
  return uint32_t(data - in);
}

// ---------------------------------------------------------------------------
// doc-block encode — WriteTailDelta (format_block_128.hpp:57-242)
// docs sorted strictly ascending, prev < docs[0], 1 <= len <= 128.
// Returns bytes written (tag byte included).
// ---------------------------------------------------------------------------
static uint32_t encode_doc_block(const uint32_t* in, uint32_t len,
                                 uint32_t prev, uint8_t* out) {
  uint8_t best_enc = SDB_DE_VALUES;
  uint32_t best_size = len * 4;

  bool all_same = true;
  const uint32_t max = in[len - 1];
  const uint32_t for_base = prev;
  const uint32_t for_max = max - for_base;

  uint32_t delta_prev = prev;
  uint32_t delta_max = in[0] - delta_prev;

  const uint32_t groups = (len + 3) / 4;
  uint32_t size_svb = 2 + groups;
  uint32_t size_delta_svb = 2 + groups;

  uint32_t deltas[kBlock];
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t value = in[i];
    const uint32_t delta_value = value - delta_prev;
    delta_prev = value;
    deltas[i] = delta_value;
    all_same &= (delta_max == delta_value);
    delta_max = std::max(delta_max, delta_value);
    size_svb += byte_size_1234(value);
    size_delta_svb += byte_size_1234(delta_value);
  }

  bool decided = false;
  if (all_same) {
    switch (byte_size_0124(delta_max)) {
      case 1: best_enc = SDB_DE_DELTA_ALL_SAME_08; best_size = 1; break;
      case 2: best_enc = SDB_DE_DELTA_ALL_SAME_16; best_size = 2; break;
      default: best_enc = SDB_DE_DELTA_ALL_SAME_32; best_size = 4; break;
    }
    decided = true;
  }
  if (!decided) {
    if (len == kBlock) {  // SupportIfBlock
      const uint32_t bits = bit_width_u32(delta_max);  // >= 2 here
      const uint32_t size = (kBlock * bits + 7) / 8;
      if (size < best_size && bits <= 31) {
        best_enc = uint8_t(SDB_DE_DELTA_BITPACK(bits));
        best_size = size;
      }
    }
    if (len != kBlock && size_svb < best_size) {  // SupportIfTail
      best_enc = SDB_DE_STREAMVBYTE1234;
      best_size = size_svb;
    }
    if (len != kBlock && size_delta_svb < best_size) {
      best_enc = SDB_DE_DELTA_STREAMVBYTE1234;
      best_size = size_delta_svb;
    }
    {
      const uint32_t words = (for_max + 1 + 63) / 64;
      const uint32_t size = 1 + words * 8;
      if (size - 2 < best_size) {
        best_enc = SDB_DE_FOR_BITSET;
        best_size = size;
      }
    }
  }

  uint8_t* p = out;
  *p++ = best_enc;
  switch (best_enc) {
    case SDB_DE_VALUES:
      std::memcpy(p, in, size_t(len) * 4);
      p += size_t(len) * 4;
      break;
    case SDB_DE_DELTA_ALL_SAME_08:
      *p++ = uint8_t(delta_max);
      break;
    case SDB_DE_DELTA_ALL_SAME_16:
      std::memcpy(p, &delta_max, 2);
      p += 2;
      break;
    case SDB_DE_DELTA_ALL_SAME_32:
      std::memcpy(p, &delta_max, 4);
      p += 4;
      break;
    case SDB_DE_FOR_BITSET: {
      // WriteBitset (format_block_128.hpp:815-839)
      const uint32_t bytes = best_size - 1;
      const uint32_t words = bytes / 8;
      uint64_t bitset[kBlock * 4 / 8];  // max 512 bytes = 64 words
      std::memset(bitset, 0, bytes);
      for (uint32_t i = 0; i < len; ++i) {
        const uint32_t v = in[i] - prev;
        bitset[v >> 6] |= 1ull << (v & 63);
      }
      *p++ = uint8_t(words);
      std::memcpy(p, bitset, bytes);
      p += bytes;
      break;
    }
    case SDB_DE_STREAMVBYTE1234: {
      uint8_t buf[kBlock * 5 + 8];
      const uint32_t size = svb_encode(in, len, buf);
      const uint16_t s16 = uint16_t(size);
      std::memcpy(p, &s16, 2);
      p += 2;
      std::memcpy(p, buf, size);
      p += size;
      break;
    }
    case SDB_DE_DELTA_STREAMVBYTE1234: {
      uint8_t buf[kBlock * 5 + 8];
      const uint32_t size = svb_encode(deltas, len, buf);
      const uint16_t s16 = uint16_t(size);
      std::memcpy(p, &s16, 2);
      p += 2;
      std::memcpy(p, buf, size);
      p += size;
      break;
    }
    default: {  // delta bitpack
      const uint32_t bits = uint32_t(best_enc - SDB_DE_DELTA_BITPACK_02) + 2;
      pack_vertical(deltas, bits, p);
      p += best_size;
      break;
    }
  }
  return uint32_t(p - out);
}

// doc-block decode — ReadTailDelta (format_block_128.hpp:466-559).
// Writes exactly `len` doc ids to out. Returns bytes consumed.
static uint32_t decode_doc_block(const uint8_t* in, uint32_t len,
                                 uint32_t prev, uint32_t* out) {
  const uint8_t* p = in;
  const uint8_t type = *p++;
  switch (type) {
    case SDB_DE_VALUES:
      std::memcpy(out, p, size_t(len) * 4);
      p += size_t(len) * 4;
      break;
    case SDB_DE_DELTA_ALL_SAME_08: {
      const uint32_t v = *p++;
      for (uint32_t i = 0; i < len; ++i) out[i] = prev + v + v * i;
      break;
    }
    case SDB_DE_DELTA_ALL_SAME_16: {
      uint16_t v16;
      std::memcpy(&v16, p, 2);
      p += 2;
      const uint32_t v = v16;
      for (uint32_t i = 0; i < len; ++i) out[i] = prev + v + v * i;
      break;
    }
    case SDB_DE_DELTA_ALL_SAME_32: {
      uint32_t v;
      std::memcpy(&v, p, 4);
      p += 4;
      for (uint32_t i = 0; i < len; ++i) out[i] = prev + v + v * i;
      break;
    }
    case SDB_DE_FOR_BITSET: {
      const uint32_t words = *p++;
      uint32_t n = 0;
      for (uint32_t i = 0; i < words; ++i) {
        uint64_t word;
        std::memcpy(&word, p + 8u * i, 8);
        const uint32_t off = prev + i * 64;
        while (word) {
          out[n++] = off + uint32_t(__builtin_ctzll(word));
          word &= word - 1;
        }
      }
      p += 8u * words;
      break;
    }
    case SDB_DE_STREAMVBYTE1234: {
      uint16_t size;
      std::memcpy(&size, p, 2);
      p += 2;
      svb_decode(p, out, len);
      p += size;
      break;
    }
    case SDB_DE_DELTA_STREAMVBYTE1234: {
      uint16_t size;
      std::memcpy(&size, p, 2);
      p += 2;
      svb_decode(p, out, len);
      p += size;
      uint32_t acc = prev;
      for (uint32_t i = 0; i < len; ++i) {
        acc += out[i];
        out[i] = acc;
      }
      break;
    }
    default: {  // delta bitpack (full blocks only)
      const uint32_t bits = uint32_t(type - SDB_DE_DELTA_BITPACK_02) + 2;
      uint32_t deltas[kBlock];
      unpack_vertical(p, bits, deltas);
      uint32_t acc = prev;
      for (uint32_t i = 0; i < kBlock; ++i) {
        acc += deltas[i];
        out[i] = acc;
      }
      p += 16u * bits;
      break;
    }
  }
  return uint32_t(p - in);
}

// freq-block encode — WriteTail (format_block_128.hpp:249-379)
static uint32_t encode_freq_block(const uint32_t* in, uint32_t len,
                                  uint8_t* out) {
  uint8_t best_enc = SDB_E_VALUES;
  uint32_t best_size = len * 4;

  bool all_same = true;
  uint32_t max = in[0];
  const uint32_t groups = (len + 3) / 4;
  uint32_t size_svb = 2 + groups;
  for (uint32_t i = 0; i < len; ++i) {
    const uint32_t value = in[i];
    all_same &= (max == value);
    max = std::max(max, value);
    size_svb += byte_size_1234(value);
  }

  bool decided = false;
  if (all_same) {
    switch (byte_size_0124(max)) {
      case 0:
      case 1: best_enc = SDB_E_ALL_SAME_08; best_size = 1; break;
      case 2: best_enc = SDB_E_ALL_SAME_16; best_size = 2; break;
      default: best_enc = SDB_E_ALL_SAME_32; best_size = 4; break;
    }
    decided = true;
  }
  if (!decided) {
    if (len == kBlock) {
      const uint32_t bits = bit_width_u32(max);  // >= 1 here
      const uint32_t size = (kBlock * bits + 7) / 8;
      if (size < best_size && bits <= 31) {
        best_enc = uint8_t(SDB_E_BITPACK(bits));
        best_size = size;
      }
    }
    if (len != kBlock && size_svb < best_size) {
      best_enc = SDB_E_STREAMVBYTE1234;
      best_size = size_svb;
    }
  }

  uint8_t* p = out;
  *p++ = best_enc;
  switch (best_enc) {
    case SDB_E_VALUES:
      std::memcpy(p, in, size_t(len) * 4);
      p += size_t(len) * 4;
      break;
    case SDB_E_ALL_SAME_08:
      *p++ = uint8_t(max);
      break;
    case SDB_E_ALL_SAME_16:
      std::memcpy(p, &max, 2);
      p += 2;
      break;
    case SDB_E_ALL_SAME_32:
      std::memcpy(p, &max, 4);
      p += 4;
      break;
    case SDB_E_STREAMVBYTE1234: {
      uint8_t buf[kBlock * 5 + 8];
      const uint32_t size = svb_encode(in, len, buf);
      const uint16_t s16 = uint16_t(size);
      std::memcpy(p, &s16, 2);
      p += 2;
      std::memcpy(p, buf, size);
      p += size;
      break;
    }
    default: {  // bitpack
      const uint32_t bits = uint32_t(best_enc - SDB_E_BITPACK_01) + 1;
      pack_vertical(in, bits, p);
      p += best_size;
      break;
    }
  }
  return uint32_t(p - out);
}

// freq-block decode — ReadTail (format_block_128.hpp:568-636)
static uint32_t decode_freq_block(const uint8_t* in, uint32_t len,
                                  uint32_t* out) {
  const uint8_t* p = in;
  const uint8_t type = *p++;
  switch (type) {
    case SDB_E_VALUES:
      std::memcpy(out, p, size_t(len) * 4);
      p += size_t(len) * 4;
      break;
    case SDB_E_ALL_SAME_08: {
      const uint32_t v = *p++;
      for (uint32_t i = 0; i < len; ++i) out[i] = v;
      break;
    }
    case SDB_E_ALL_SAME_16: {
      uint16_t v;
      std::memcpy(&v, p, 2);
      p += 2;
      for (uint32_t i = 0; i < len; ++i) out[i] = v;
      break;
    }
    case SDB_E_ALL_SAME_32: {
      uint32_t v;
      std::memcpy(&v, p, 4);
      p += 4;
      for (uint32_t i = 0; i < len; ++i) out[i] = v;
      break;
    }
    case SDB_E_STREAMVBYTE1234: {
      uint16_t size;
      std::memcpy(&size, p, 2);
      p += 2;
      svb_decode(p, out, len);
      p += size;
      break;
    }
    default: {  // bitpack (full blocks only)
      const uint32_t bits = uint32_t(type - SDB_E_BITPACK_01) + 1;
      unpack_vertical(p, bits, out);
      p += 16u * bits;
      break;
    }
  }
  return uint32_t(p - in);
}

// ---------------------------------------------------------------------------
// FoR/bitpack i64 column codec — this repo's own
// layout (the reference's column codecs live in the un-vendored DuckDB
// fork; parity is at result level). Per row group:
// frame-of-reference base (min) + horizontal bitpack of (v - base) at the
// group's required width, plus a min/max zonemap (BaseStatistics analogue,
// column_reader.hpp:201) for whole-group predicate skips (DeadUntil,
// full_scanner.h:61-71).
//
// Blob layout (64-bit aligned):
//   SdbColHeader { u64 magic='SDBCOL01'; u64 rows; u32 group_rows;
//                  u32 ngroups; u64 off_desc; u64 off_payload; u64 size; }
//   SdbColGroupDesc[ngroups] { i64 base; i64 vmin; i64 vmax; u64 bit_off;
//                              u16 width; u16 pad[3]; }
//   payload: per group, ceil(group_rows*width/32) u32 words, packed
//   little-endian, value i at bits [i*width, (i+1)*width).
// ---------------------------------------------------------------------------
extern "C" {
typedef struct SdbColHeader {
  uint64_t magic;
  uint64_t rows;
  uint32_t group_rows;
  uint32_t ngroups;
  uint64_t off_desc;
  uint64_t off_payload;
  uint64_t size;
} SdbColHeader;
typedef struct SdbColGroupDesc {
  int64_t base;
  int64_t vmin;
  int64_t vmax;
  uint64_t word_off; /* u32-word offset into payload */
  uint16_t width;    /* bits per value, 0..32 (delta range must fit u32) */
  uint16_t pad[3];
} SdbColGroupDesc;
}
#define SDB_COL_MAGIC 0x31304C4F43424453ull

extern "C" int sdb_host_encode_col_i64(const int64_t* vals, uint64_t rows,
                                       uint32_t group_rows, void** blob_out,
                                       uint64_t* blob_size) {
  if (!vals || !blob_out || !blob_size || rows == 0 || group_rows == 0)
    return -1;
  const uint32_t ngroups = uint32_t((rows + group_rows - 1) / group_rows);
  std::vector<SdbColGroupDesc> desc(ngroups);
  std::vector<uint32_t> payload;
  payload.reserve(rows / 2);
  for (uint32_t g = 0; g < ngroups; ++g) {
    const uint64_t r0 = uint64_t(g) * group_rows;
    const uint64_t r1 = std::min<uint64_t>(rows, r0 + group_rows);
    int64_t mn = vals[r0], mx = vals[r0];
    for (uint64_t r = r0; r < r1; ++r) {
      mn = std::min(mn, vals[r]);
      mx = std::max(mx, vals[r]);
    }
    const uint64_t range = uint64_t(mx - mn);
    if (range > 0xFFFFFFFFull) return -2; /* delta must fit u32 (round 1) */
    uint32_t width = 0;
    while (width < 33 && (width == 64 ? 0 : (range >> width)))
      ++width; /* bit_width(range) */
    SdbColGroupDesc d{};
    d.base = mn;
    d.vmin = mn;
    d.vmax = mx;
    d.width = uint16_t(width);
    d.word_off = payload.size();
    const uint64_t nw = (uint64_t(r1 - r0) * width + 31) / 32;
    const size_t p0 = payload.size();
    payload.resize(p0 + nw, 0u);
    // width == 0 (constant group): nw == 0, nothing to pack — the old
    // unconditional loop wrote payload[p0] |= 0 one past the end (a
    // null-deref for tiny columns; found by tests/test_codec_fuzz.py)
    if (width)
      for (uint64_t r = r0; r < r1; ++r) {
        const uint64_t v = uint64_t(vals[r] - mn);
        const uint64_t bit = (r - r0) * width;
        const uint64_t w = bit >> 5;
        const uint32_t sh = uint32_t(bit & 31);
        payload[p0 + w] |= uint32_t(v << sh);
        if (sh + width > 32) payload[p0 + w + 1] |= uint32_t(v >> (32 - sh));
      }
    desc[g] = d;
  }
  auto align64 = [](uint64_t x) { return (x + 63) & ~63ull; };
  SdbColHeader hdr{};
  hdr.magic = SDB_COL_MAGIC;
  hdr.rows = rows;
  hdr.group_rows = group_rows;
  hdr.ngroups = ngroups;
  hdr.off_desc = align64(sizeof(SdbColHeader));
  hdr.off_payload = align64(hdr.off_desc + sizeof(SdbColGroupDesc) * ngroups);
  hdr.size = align64(hdr.off_payload + payload.size() * 4 + 64);
  uint8_t* blob = static_cast<uint8_t*>(std::calloc(1, hdr.size));
  if (!blob) return -4;
  std::memcpy(blob, &hdr, sizeof(hdr));
  std::memcpy(blob + hdr.off_desc, desc.data(),
              sizeof(SdbColGroupDesc) * ngroups);
  if (!payload.empty())
    std::memcpy(blob + hdr.off_payload, payload.data(), payload.size() * 4);
  *blob_out = blob;
  *blob_size = hdr.size;
  return 0;
}

// scalar decode (builder verification + oracle cross-check)
extern "C" int sdb_host_decode_col_i64(const void* blob, uint64_t size,
                                       int64_t* out, uint64_t rows) {
  if (!blob || size < sizeof(SdbColHeader)) return -5;
  const SdbColHeader* hdr = (const SdbColHeader*)blob;
  if (hdr->magic != SDB_COL_MAGIC || hdr->rows != rows || hdr->size > size ||
      hdr->group_rows == 0)
    return -5;
  // every offset/extent must sit inside the declared blob span (a truncated
  // or corrupted blob must fail loudly, not read out of bounds)
  const uint64_t ng_need = (rows + hdr->group_rows - 1) / hdr->group_rows;
  if (hdr->ngroups != ng_need ||
      hdr->off_desc > hdr->size || hdr->off_payload > hdr->size ||
      uint64_t(hdr->ngroups) * sizeof(SdbColGroupDesc) >
        hdr->size - hdr->off_desc)
    return -5;
  const uint64_t paywords = (hdr->size - hdr->off_payload) / 4;
  const auto* desc =
    (const SdbColGroupDesc*)((const uint8_t*)blob + hdr->off_desc);
  const auto* pl = (const uint32_t*)((const uint8_t*)blob + hdr->off_payload);
  for (uint32_t g = 0; g < hdr->ngroups; ++g) {
    const uint64_t r0 = uint64_t(g) * hdr->group_rows;
    const uint64_t r1 = std::min<uint64_t>(rows, r0 + hdr->group_rows);
    const SdbColGroupDesc d = desc[g];
    if (d.width > 32) return -5;
    if (d.width &&
        (d.word_off > paywords ||
         ((r1 - r0) * d.width + 31) / 32 > paywords - d.word_off))
      return -5;
    const uint32_t* w = pl + d.word_off;
    const uint64_t mask =
      d.width >= 32 ? 0xFFFFFFFFull : ((1ull << d.width) - 1);
    for (uint64_t r = r0; r < r1; ++r) {
      if (d.width == 0) {
        out[r] = d.base;
        continue;
      }
      const uint64_t bit = (r - r0) * d.width;
      uint64_t v = w[bit >> 5] >> (bit & 31);
      if ((bit & 31) + d.width > 32)
        v |= uint64_t(w[(bit >> 5) + 1]) << (32 - (bit & 31));
      out[r] = d.base + int64_t(v & mask);
    }
  }
  return 0;
}

// ---------------------------------------------------------------------------
// seeded synthetic corpus: stateless splitmix64 hashing so
// any (term, doc) draw is order-independent and parallelizable.
// ---------------------------------------------------------------------------
inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
inline uint64_t hash3(uint64_t seed, uint64_t a, uint64_t b) {
  return splitmix64(seed ^ splitmix64(a * 0x9E3779B97F4A7C15ull ^
                                      splitmix64(b + 0xD1B54A32D192ED03ull)));
}
inline double u01(uint64_t h) {  // (0,1)
  return (double(h >> 11) + 0.5) * (1.0 / 9007199254740992.0);
}

// freq ~ shifted geometric(p=0.6) capped at 255: P(n) = p(1-p)^(n-1), n>=1
static uint32_t synth_freq(uint64_t seed, uint32_t term, uint32_t doc) {
  const double u = u01(hash3(seed, 0x66726571u ^ term, doc));
  const int n = 1 + int(std::log(u) / std::log(0.4));
  return uint32_t(std::min(255, std::max(1, n)));
}

// norm ~ round(lognormal(mu=ln 120, sigma=0.5)), >= 1 (Box–Muller)
static uint32_t synth_norm(uint64_t seed, uint32_t doc) {
  const double u1 = u01(hash3(seed, 0x6E6F726Du, doc));
  const double u2 = u01(hash3(seed, 0x6E6F726Eu, doc));
  const double z =
    std::sqrt(-2.0 * std::log(u1)) * std::cos(2.0 * M_PI * u2);
  const double v = std::exp(std::log(120.0) + 0.5 * z);
  return uint32_t(std::max(1.0, std::min(4.0e9, std::round(v))));
}

static bool synth_member(uint64_t seed, uint32_t term, uint32_t doc,
                         double sel) {
  return u01(hash3(seed, 0x6D656D62u ^ (uint64_t(term) << 32), doc)) < sel;
}

}  // namespace

// ---------------------------------------------------------------------------
// exported C ABI (host side)
// ---------------------------------------------------------------------------
extern "C" {

int sdb_host_encode_doc_block(const uint32_t* docs, uint32_t len,
                              uint32_t prev, uint8_t* out,
                              uint32_t* out_size) {
  if (!docs || !out || !out_size || len < 1 || len > kBlock) return -1;
  *out_size = encode_doc_block(docs, len, prev, out);
  return 0;
}
int sdb_host_decode_doc_block(const uint8_t* in, uint32_t len, uint32_t prev,
                              uint32_t* out, uint32_t* consumed) {
  if (!in || !out || len < 1 || len > kBlock) return -1;
  const uint32_t c = decode_doc_block(in, len, prev, out);
  if (consumed) *consumed = c;
  return 0;
}
int sdb_host_encode_freq_block(const uint32_t* freqs, uint32_t len,
                               uint8_t* out, uint32_t* out_size) {
  if (!freqs || !out || !out_size || len < 1 || len > kBlock) return -1;
  *out_size = encode_freq_block(freqs, len, out);
  return 0;
}
int sdb_host_decode_freq_block(const uint8_t* in, uint32_t len, uint32_t* out,
                               uint32_t* consumed) {
  if (!in || !out || len < 1 || len > kBlock) return -1;
  const uint32_t c = decode_freq_block(in, len, out);
  if (consumed) *consumed = c;
  return 0;
}

// Synthetic corpus: per-term postings of term `t` under (seed, sel).
// If docs==nullptr, only counts df. docs/freqs sized >= df.
int sdb_host_synth_postings(uint64_t seed, uint32_t doc_count, uint32_t term,
                            double sel, uint32_t* docs, uint32_t* freqs,
                            uint32_t* df_out) {
  uint32_t n = 0;
  for (uint32_t d = 1; d <= doc_count; ++d) {
    if (synth_member(seed, term, d, sel)) {
      if (docs) {
        docs[n] = d;
        freqs[n] = synth_freq(seed, term, d);
      }
      ++n;
    }
  }
  if (df_out) *df_out = n;
  return 0;
}

int sdb_host_synth_norms(uint64_t seed, uint32_t doc_count, uint32_t* norms) {
  for (uint32_t d = 1; d <= doc_count; ++d) norms[d] = synth_norm(seed, d);
  norms[0] = 0;
  return 0;
}

// Build a serialized segment blob from explicit postings. Arrays:
//   df[t], docs[t][0..df[t]), freqs[t][...], norms[1..doc_count].
// Caller frees *blob with sdb_host_blob_free.
int sdb_host_build_segment(uint32_t doc_count, uint32_t nterms,
                           const uint32_t* df, const uint32_t* const* docs,
                           const uint32_t* const* freqs, const uint32_t* norms,
                           void** blob_out, uint64_t* blob_size) {
  if (!df || !blob_out || !blob_size) return -1;

  std::vector<SdbTermEntry> terms(nterms);
  std::vector<SdbBlockDesc> desc;
  std::vector<uint8_t> payload;
  payload.reserve(1 << 20);

  uint64_t total_tf = 0;
  for (uint32_t t = 0; t < nterms; ++t) {
    auto& te = terms[t];
    te.desc_begin = desc.size();
    te.payload_begin = payload.size();
    te.df = df[t];
    te.max_freq = 0;
    te.total_freq = 0;
    const uint32_t* td = df[t] ? docs[t] : nullptr;
    const uint32_t* tf = df[t] ? freqs[t] : nullptr;
    uint32_t prev = 0;  // doc ids start at 1; initial delta base is 0
    for (uint32_t pos = 0; pos < df[t]; pos += kBlock) {
      const uint32_t len = std::min(kBlock, df[t] - pos);
      SdbBlockDesc bd{};
      bd.prev_doc = prev;
      bd.last_doc = td[pos + len - 1];
      bd.len = uint16_t(len);
      bd.doc_off = uint32_t(payload.size() - te.payload_begin);
      uint8_t buf[kBlock * 5 + 16];
      uint32_t sz = encode_doc_block(td + pos, len, prev, buf);
      payload.insert(payload.end(), buf, buf + sz);
      bd.freq_off = uint32_t(payload.size() - te.payload_begin);
      sz = encode_freq_block(tf + pos, len, buf);
      payload.insert(payload.end(), buf, buf + sz);
      const uint8_t ftag = buf[0];
      /* v2: per-block norm stream right after the freq block, encoded with
       * the same non-delta families. This materializes the norm-column
       * gather at index build time: query results are
       * identical (parity tests compare against the column-reading oracle
       * bit-for-bit), HBM traffic drops from a scattered 4 B/posting
       * gather to ~1.5 B/posting of sequential payload. */
      uint8_t ntag;
      {
        uint32_t nvals[kBlock];
        for (uint32_t i = 0; i < len; ++i)
          nvals[i] = norms ? norms[td[pos + i]] : 1u;
        const uint32_t nsz = encode_freq_block(nvals, len, buf);
        payload.insert(payload.end(), buf, buf + nsz);
        ntag = buf[0];
      }
      /* flags (sdb_format.h): for the fused shape carry all three bit
       * widths so the GPU decode needs no payload-tag fetch before
       * issuing its packed-word loads; otherwise carry the freq-block
       * size (norm_off derivation) with bit0 clear */
      const uint8_t dtag2 = payload[te.payload_begin + bd.doc_off];
      if (len == kBlock && dtag2 >= SDB_DE_DELTA_BITPACK_02 &&
          ftag >= SDB_E_BITPACK_01 && ntag >= SDB_E_BITPACK_01) {
        const uint32_t dbits = dtag2 - SDB_DE_DELTA_BITPACK_02 + 2;
        const uint32_t fbits = ftag - SDB_E_BITPACK_01 + 1;
        const uint32_t nbits = ntag - SDB_E_BITPACK_01 + 1;
        bd.flags = uint16_t(1u | (dbits << 1) | (fbits << 6) |
                            (nbits << 11));
      } else {
        bd.flags = uint16_t(sz << 1);
      }
      uint32_t mf = 0, mn = 0xFFFFFFFFu;
      for (uint32_t i = 0; i < len; ++i) {
        mf = std::max(mf, tf[pos + i]);
        te.total_freq += tf[pos + i];
        if (norms) mn = std::min(mn, norms[td[pos + i]]);
      }
      bd.max_freq = mf;
      bd.min_norm = mn;
      te.max_freq = std::max(te.max_freq, mf);
      desc.push_back(bd);
      prev = bd.last_doc;
    }
    te.desc_end = desc.size();
    te.payload_end = payload.size();
  }
  for (uint32_t d = 1; d <= doc_count; ++d) total_tf += norms ? norms[d] : 1;

  auto align64 = [](uint64_t x) { return (x + 63) & ~63ull; };
  SdbSegHeader hdr{};
  hdr.magic = SDB_SEG_MAGIC;
  hdr.version = 3; /* v3 = per-block norm streams + width-carrying
                  * descriptor flags (sdb_format.h) */
  hdr.nterms = nterms;
  hdr.doc_count = doc_count;
  hdr.docs_with_field = doc_count;
  hdr.total_term_freq = total_tf;
  hdr.total_blocks = desc.size();
  uint64_t off = align64(sizeof(SdbSegHeader));
  hdr.off_terms = off;
  off = align64(off + sizeof(SdbTermEntry) * nterms);
  hdr.off_desc = off;
  off = align64(off + sizeof(SdbBlockDesc) * desc.size());
  hdr.off_norms = off;
  off = align64(off + sizeof(uint32_t) * (uint64_t(doc_count) + 1));
  hdr.off_payload = off;
  hdr.payload_size = payload.size();
  /* +64 tail pad: device decode reads payload words via aligned-pair loads
   * that may touch up to 7 bytes past the last used byte */
  hdr.blob_size = align64(off + payload.size() + 64);

  uint8_t* blob = static_cast<uint8_t*>(std::calloc(1, hdr.blob_size));
  if (!blob) return -4;
  std::memcpy(blob, &hdr, sizeof(hdr));
  std::memcpy(blob + hdr.off_terms, terms.data(),
              sizeof(SdbTermEntry) * nterms);
  if (!desc.empty())
    std::memcpy(blob + hdr.off_desc, desc.data(),
                sizeof(SdbBlockDesc) * desc.size());
  auto* nb = reinterpret_cast<uint32_t*>(blob + hdr.off_norms);
  if (norms)
    std::memcpy(nb, norms, sizeof(uint32_t) * (uint64_t(doc_count) + 1));
  else
    for (uint32_t d = 0; d <= doc_count; ++d) nb[d] = 1;
  if (!payload.empty())
    std::memcpy(blob + hdr.off_payload, payload.data(), payload.size());

  *blob_out = blob;
  *blob_size = hdr.blob_size;
  return 0;
}

// Convenience: build the whole synthetic segment for docs [doc_lo, doc_hi]
// of a (seed, selectivities) corpus — doc ids are kept GLOBAL so sharded
// segments on N ranks score identically to one big segment.
// NOTE: field stats written into the blob cover ONLY this shard; global BM25
// stats for sharded execution are supplied via sdb_host_bm25_stats inputs.
int sdb_host_build_synth_segment(uint64_t seed, uint32_t doc_lo,
                                 uint32_t doc_hi, uint32_t nterms,
                                 const double* sel, void** blob_out,
                                 uint64_t* blob_size) {
  if (doc_lo < 1 || doc_hi < doc_lo) return -1;
  const uint32_t n_docs = doc_hi - doc_lo + 1;
  std::vector<std::vector<uint32_t>> docs(nterms), freqs(nterms);
  for (uint32_t t = 0; t < nterms; ++t) {
    for (uint32_t d = doc_lo; d <= doc_hi; ++d) {
      if (synth_member(seed, t, d, sel[t])) {
        docs[t].push_back(d);
        freqs[t].push_back(synth_freq(seed, t, d));
      }
    }
  }
  // norms indexed by LOCAL offset +1? No: global doc ids -> segment stores
  // norms for [doc_lo-1 .. doc_hi] at positions [doc_lo-1 .. doc_hi]; to keep
  // the blob dense we build with doc ids SHIFTED to 1..n_docs and record the
  // shard base separately at query time. Shift here:
  std::vector<uint32_t> norms(size_t(n_docs) + 1);
  norms[0] = 0;
  for (uint32_t d = 0; d < n_docs; ++d)
    norms[d + 1] = synth_norm(seed, doc_lo + d);
  std::vector<const uint32_t*> dp(nterms), fp(nterms);
  std::vector<uint32_t> df(nterms);
  std::vector<std::vector<uint32_t>> sdocs(nterms);
  for (uint32_t t = 0; t < nterms; ++t) {
    df[t] = uint32_t(docs[t].size());
    sdocs[t].resize(docs[t].size());
    for (size_t i = 0; i < docs[t].size(); ++i)
      sdocs[t][i] = docs[t][i] - (doc_lo - 1);
    dp[t] = sdocs[t].data();
    fp[t] = freqs[t].data();
  }
  return sdb_host_build_segment(n_docs, nterms, df.data(), dp.data(),
                                fp.data(), norms.data(), blob_out, blob_size);
}

void sdb_host_blob_free(void* blob) { std::free(blob); }

int sdb_host_segment_parse(const void* blob, uint64_t size,
                           SdbSegmentView* out) {
  if (!blob || size < sizeof(SdbSegHeader)) return -5;
  const auto* hdr = static_cast<const SdbSegHeader*>(blob);
  if (hdr->magic != SDB_SEG_MAGIC || hdr->version < 1 ||
      hdr->version > 3 || hdr->version == 2 /* retired flags layout */ ||
      hdr->blob_size > size)
    return -5;
  const auto* base = static_cast<const uint8_t*>(blob);
  out->hdr = hdr;
  out->terms = reinterpret_cast<const SdbTermEntry*>(base + hdr->off_terms);
  out->desc = reinterpret_cast<const SdbBlockDesc*>(base + hdr->off_desc);
  out->norms = reinterpret_cast<const uint32_t*>(base + hdr->off_norms);
  out->payload = base + hdr->off_payload;
  return 0;
}

// BM25 stats — BM25::collect (search/bm25.cpp:279-306): idf accumulated via
// double log1p narrowed to f32; norm_const = k(1-b); norm_length = k*b/avgDL
// with avgDL = (f32)total_term_freq / (f32)docs_with_field.
void sdb_host_bm25_stats(uint64_t docs_with_field, uint64_t docs_with_term,
                         uint64_t total_term_freq, float k, float b,
                         float* idf, float* norm_const, float* norm_length) {
  *idf = float(std::log1p(
    (double(docs_with_field - docs_with_term) + 0.5) /
    (double(docs_with_term) + 0.5)));
  const float kb = k * b;
  if (b == 0.0f) {  // BM15: stats->norm_const = k (bm25.cpp:296-299)
    *norm_const = k;
    *norm_length = 0.0f;
    return;
  }
  *norm_const = k - kb;
  if (total_term_freq && docs_with_field) {
    const float avg_dl = float(total_term_freq) / float(docs_with_field);
    *norm_length = kb / avg_dl;
  } else {
    *norm_length = kb;
  }
}

// Final host-side top-k select over candidate (score,doc,segment) triples —
// the PrepareEmitBuffer analogue (duckdb_search_full_scan.cpp:1945-2000) with
// the deterministic tie order of DESIGN.md: (score desc, segment asc, doc
// asc). Acceptance threshold mirrors doc_collector.hpp:58: score > FLT_MIN.
typedef struct SdbScoreDocC {
  float score;
  uint32_t doc;
  uint32_t segment_idx;
} SdbScoreDocC;

int sdb_host_topk_select(const SdbScoreDocC* cands, uint64_t n, uint32_t k,
                         SdbScoreDocC* out, uint32_t* out_count) {
  std::vector<SdbScoreDocC> v;
  v.reserve(n);
  const float kFltMin = 1.17549435e-38f;  // std::numeric_limits<float>::min()
  for (uint64_t i = 0; i < n; ++i)
    if (cands[i].score > kFltMin) v.push_back(cands[i]);
  auto cmp = [](const SdbScoreDocC& a, const SdbScoreDocC& b) {
    if (a.score != b.score) return a.score > b.score;
    if (a.segment_idx != b.segment_idx) return a.segment_idx < b.segment_idx;
    return a.doc < b.doc;
  };
  const size_t kk = std::min<size_t>(k, v.size());
  std::partial_sort(v.begin(), v.begin() + kk, v.end(), cmp);
  std::copy(v.begin(), v.begin() + kk, out);
  *out_count = uint32_t(kk);
  return 0;
}

}  // extern "C"


/* ------------------------------------------------------------------------- */
/* On-disk reference ingestion: read the reference's   */
/* `.doc` postings stream (PostingsWriterBase, formats/posting/writer.hpp)   */
/* and rebuild it as this project's segment container. The term metadata the */
/* burst-trie terms dict (.tm) would supply arrives as a caller sidecar      */
/* (SdbDocTermMeta == the TermMetaImpl fields Encode() serializes,           */
/* writer.hpp:379-412); the burst-trie itself is SURVEY.md §2 out of scope   */
/* this round.                                                               */
/*                                                                           */
/* File layout restated from the reference:                                  */
/*  header  : u32 LE magic 0x3fd76c17, vint-length + "iresearch_10_         */
/*            postings_documents", u32 version (format_utils.cpp:56-64)      */
/*  per term (TermMeta doc_start): full 128-doc blocks [WriteBlockDelta      */
/*            docs][WriteBlock freqs] (writer.hpp:618-627); then            */
/*            df==1        -> nothing (e_single_doc in meta)                 */
/*            df<=128      -> [root wand: u8 size + bytes][tail delta docs   */
/*                            + tail freqs]  (EndTerm writer.hpp:346-369:    */
/*                            write_max_score(0) BEFORE FlushTailDoc)        */
/*            df>128       -> [tail...][@e_skip_start: root wand][v32        */
/*                            num_levels][levels n..0: v64 len + bytes]      */
/*            level-0 entry: v32 absolute block-last doc, v64 doc-ptr delta  */
/*            (base doc_start), u8 wand size, wand bytes (vint freq [+ vint  */
/*            norm-freq], wand_writer.hpp:196-206); level>=1 entries append  */
/*            v64 child position (skip_list.hpp:92-116)                      */
/*  footer  : u32 LE -magic, u32 alg 0, u64 CRC-32C of file[0..len-8)        */
/*            (format_utils.cpp WriteFooter; libs/basics/crc.hpp = absl      */
/*            crc32c = Castagnoli, reflected, init/final-xor 0xffffffff)     */
/* ------------------------------------------------------------------------- */

static uint32_t g_crc32c_tab[256];
static bool g_crc32c_ready = false;
static void crc32c_build_tab() {
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = i;
    for (int k = 0; k < 8; ++k)
      c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
    g_crc32c_tab[i] = c;
  }
  g_crc32c_ready = true;
}
static uint32_t crc32c_bytes(const uint8_t* p, size_t n) {
  if (!g_crc32c_ready) crc32c_build_tab();
  uint32_t c = 0xFFFFFFFFu;
  for (size_t i = 0; i < n; ++i)
    c = g_crc32c_tab[(c ^ p[i]) & 0xFFu] ^ (c >> 8);
  return c ^ 0xFFFFFFFFu;
}

struct DocCursor {
  const uint8_t* p;
  const uint8_t* end;
  bool fail = false;
  uint8_t u8() {
    if (p >= end) { fail = true; return 0; }
    return *p++;
  }
  uint32_t u32le() {
    if (end - p < 4) { fail = true; return 0; }
    uint32_t v;
    std::memcpy(&v, p, 4);
    p += 4;
    return v;  /* LE host assumed (x86/gfx hosts) */
  }
  uint64_t vu64() {  /* WriteVarint, data_output.hpp:58 (LEB128) */
    uint64_t v = 0;
    for (uint32_t sh = 0; sh < 64; sh += 7) {
      const uint8_t b = u8();
      v |= (uint64_t)(b & 0x7F) << sh;
      if (!(b & 0x80)) return v;
    }
    fail = true;
    return 0;
  }
  void skip(uint64_t n) {
    if ((uint64_t)(end - p) < n) { fail = true; return; }
    p += n;
  }
};

extern "C" {

int sdb_host_ingest_doc(const void* file, uint64_t size,
                        const SdbDocTermMeta* metas, uint32_t nterms,
                        uint32_t doc_count, uint32_t has_freq,
                        const uint32_t* norms /* doc_count+1 */,
                        void** blob_out, uint64_t* size_out) {
  if (!file || !metas || !blob_out || !size_out || nterms == 0 ||
      doc_count == 0)
    return -1;
  const uint8_t* f = (const uint8_t*)file;
  static const char kFmt[] = "iresearch_10_postings_documents";
  const uint64_t fmt_len = sizeof(kFmt) - 1;
  const uint64_t hdr_len = 4 + 1 + fmt_len + 4; /* vint(31) = 1 byte */
  if (size < hdr_len + 16) return -71;
  /* header */
  DocCursor h{f, f + size};
  if (h.u32le() != 0x3fd76c17u) return -72;
  if (h.vu64() != fmt_len) return -73;
  if (std::memcmp(h.p, kFmt, fmt_len) != 0) return -73;
  h.skip(fmt_len);
  (void)h.u32le(); /* version: recorded formats accept their own range */
  /* footer (format_utils.cpp ValidateFooter + WriteFooter) */
  {
    DocCursor t{f + size - 16, f + size};
    if (t.u32le() != (uint32_t)(-(int32_t)0x3fd76c17)) return -74;
    if (t.u32le() != 0) return -75;
    uint64_t want;
    std::memcpy(&want, f + size - 8, 8);
    const uint32_t got = crc32c_bytes(f, size - 8);
    if (want != (uint64_t)got) return -76;
  }

  /* decode every term's postings */
  std::vector<std::vector<uint32_t>> tdocs(nterms), tfreqs(nterms);
  for (uint32_t t = 0; t < nterms; ++t) {
    const SdbDocTermMeta& m = metas[t];
    const uint32_t df = m.docs_count;
    if (df == 0) continue;
    tdocs[t].reserve(df);
    tfreqs[t].reserve(df);
    if (df == 1) {
      const uint32_t doc = 1u + m.e_single_doc; /* + doc_limits::min() */
      if (doc > doc_count) return -77;
      tdocs[t].push_back(doc);
      tfreqs[t].push_back(has_freq ? (uint32_t)m.total_freq : 1u);
      continue;
    }
    if (m.doc_start >= size - 16) return -77;
    DocCursor c{f + m.doc_start, f + size - 16};
    const uint32_t full_blocks = df / 128u;
    const uint32_t tail_len = df % 128u;
    const bool has_skip = df > 128u;
    uint32_t prev = 0;
    uint32_t buf_d[128], buf_f[128];
    /* a block is at most 1 tag + 128*4 values + svb header slack; decode
     * through a zero-padded copy near the end of the span so a corrupt
     * length can never read past the caller's buffer */
    uint8_t safe[704];
    auto safe_docs = [&](DocCursor& cc, uint32_t len, uint32_t pv,
                         uint32_t* out) -> bool {
      const uint64_t rem = (uint64_t)(cc.end - cc.p);
      uint32_t consumed;
      if (rem >= sizeof(safe)) {
        consumed = decode_doc_block(cc.p, len, pv, out);
      } else {
        std::memset(safe, 0, sizeof(safe));
        std::memcpy(safe, cc.p, rem);
        consumed = decode_doc_block(safe, len, pv, out);
        if (consumed > rem) return false;
      }
      if (!consumed) return false;
      cc.skip(consumed);
      return !cc.fail;
    };
    auto safe_freqs = [&](DocCursor& cc, uint32_t len,
                          uint32_t* out) -> bool {
      const uint64_t rem = (uint64_t)(cc.end - cc.p);
      uint32_t consumed;
      if (rem >= sizeof(safe)) {
        consumed = decode_freq_block(cc.p, len, out);
      } else {
        std::memset(safe, 0, sizeof(safe));
        std::memcpy(safe, cc.p, rem);
        consumed = decode_freq_block(safe, len, out);
        if (consumed > rem) return false;
      }
      if (!consumed) return false;
      cc.skip(consumed);
      return !cc.fail;
    };
    auto push_block = [&](uint32_t len) -> bool {
      uint32_t last = prev;
      for (uint32_t i = 0; i < len; ++i) {
        if (buf_d[i] <= last || buf_d[i] > doc_count) return false;
        last = buf_d[i];
        tdocs[t].push_back(buf_d[i]);
        tfreqs[t].push_back(has_freq ? buf_f[i] : 1u);
      }
      prev = last;
      return true;
    };
    std::vector<uint64_t> block_end_off(full_blocks);
    for (uint32_t b = 0; b < full_blocks; ++b) {
      if (!safe_docs(c, 128, prev, buf_d)) return -78;
      if (has_freq && !safe_freqs(c, 128, buf_f)) return -78;
      if (!push_block(128)) return -79;
      block_end_off[b] = (uint64_t)(c.p - (f + m.doc_start));
    }
    auto read_wand_entry = [&](DocCursor& cc) {
      const uint8_t sz = cc.u8();
      cc.skip(sz);
    };
    if (!has_skip) {
      read_wand_entry(c); /* root wand BEFORE the tail (EndTerm order) */
      if (tail_len) {
        if (!safe_docs(c, tail_len, prev, buf_d)) return -78;
        if (has_freq && !safe_freqs(c, tail_len, buf_f)) return -78;
        if (!push_block(tail_len)) return -79;
      }
    } else {
      if (tail_len) {
        if (!safe_docs(c, tail_len, prev, buf_d)) return -78;
        if (has_freq && !safe_freqs(c, tail_len, buf_f)) return -78;
        if (!push_block(tail_len)) return -79;
      }
      /* the tail must end exactly where the meta says the skip area is */
      if ((uint64_t)(c.p - (f + m.doc_start)) != m.e_skip_start) return -80;
      DocCursor sk{f + m.doc_start + m.e_skip_start, f + size - 16};
      read_wand_entry(sk); /* root */
      const uint64_t num_levels = sk.vu64();
      if (num_levels == 0 || num_levels > 10) return -81;
      /* levels are written n..0; walk down to level 0 and validate it */
      const uint8_t* lvl0 = nullptr;
      uint64_t lvl0_len = 0;
      for (uint64_t l = 0; l < num_levels; ++l) {
        const uint64_t len = sk.vu64();
        if (sk.fail) return -81;
        if (l + 1 == num_levels) { /* last written = level 0 */
          lvl0 = sk.p;
          lvl0_len = len;
        }
        sk.skip(len);
      }
      if (sk.fail || !lvl0) return -81;
      /* level-0 entries: one per full block except the last flush without
       * a following doc (writer.hpp:733 skip cadence; no terminal skip) */
      const uint32_t nentries = tail_len ? full_blocks : full_blocks - 1;
      DocCursor e{lvl0, lvl0 + lvl0_len};
      uint64_t ptr = 0;
      for (uint32_t i = 0; i < nentries; ++i) {
        const uint32_t entry_doc = (uint32_t)e.vu64();
        ptr += e.vu64();
        read_wand_entry(e);
        if (e.fail) return -82;
        /* cross-checks against the decoded blocks */
        if (entry_doc != tdocs[t][(size_t)(i + 1) * 128 - 1]) return -83;
        if (ptr != block_end_off[i]) return -84;
      }
    }
    if (tdocs[t].size() != df) return -85;
  }

  /* rebuild as this project's container (norm streams, descriptors, WAND
   * bounds recomputed exactly as the synthetic builder does) */
  std::vector<uint32_t> df(nterms);
  std::vector<const uint32_t*> dptr(nterms), fptr(nterms);
  for (uint32_t t = 0; t < nterms; ++t) {
    df[t] = (uint32_t)tdocs[t].size();
    dptr[t] = tdocs[t].data();
    fptr[t] = tfreqs[t].data();
  }
  return sdb_host_build_segment(doc_count, nterms, df.data(),
                                (const uint32_t**)dptr.data(),
                                (const uint32_t**)fptr.data(), norms,
                                blob_out, size_out);
}

}  /* extern "C" */
