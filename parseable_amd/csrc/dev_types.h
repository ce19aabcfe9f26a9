// Shared host/device types for the gpuq execution kernels (gfx950).
#pragma once
#include <cstdint>

namespace gpuq {

// One parquet page staged in device memory.
// raw bytes at d_raw + src_off; decompressed image at d_dec + dst_off.
struct DevPage {
  uint64_t src_off;
  uint64_t dst_off;
  uint32_t comp_size;
  uint32_t uncomp_size;
  uint32_t num_values;   // rows incl. nulls (data pages)
  uint32_t row_start;    // partition-global row index of first row
  uint32_t aux;          // index into remap pool (gid decode)
  uint32_t aux_val;      // index into dict-value pool (value/rank decode) —
                         // separate from aux: a utf8 column can need BOTH
                         // (group key or count -> gid; min/max -> ranks)
  uint32_t aux_lut;      // index into predicate-LUT pool (dict-mask decode)
  uint32_t dict_n;       // dictionary entry count of this chunk (0 if none)
  uint8_t  optional;     // has def levels (max_def_level == 1)
  uint8_t  raw_copy;     // stored uncompressed: memcpy instead of LZ4
  uint8_t  encoding;     // ENC_* (meta.h)
  uint8_t  phys;         // PT_* (meta.h)
};

// LZ4 segment decompression (host-planned; meta.h Lz4Plan)
struct DevSeg {
  uint64_t src_off;    // absolute into d_raw
  uint64_t dst_off;    // absolute into d_dec
  uint32_t comp_len;
  uint32_t out_len;
  uint8_t big;         // giant single sequence: stream straight to global
  uint8_t raw;         // stored uncompressed: plain copy
};
struct DevBr {         // deferred match, absolute into d_dec (src < dst)
  uint64_t dst, src;
  uint32_t len;
  uint32_t _pad;
};
struct DevPageBr { uint32_t start, count; };  // per page: range into the br array
struct DevBrRes {      // literal-resolved record (absolute dec offsets)
  uint64_t dst;
  uint32_t len, off;
  uint32_t piece_start, piece_n;
};
struct DevPiece { uint64_t src; uint32_t len; uint32_t _pad; };
// litpar literal copy: raw[src..src+len) -> dec[dst..dst+len), PACKED to
// 16 B — at 1 B-row scale these record streams are the largest kernel
// fetch (PMC: 6-8 GB/launch at 24 B each), so the layout is the traffic.
// a = src | len << 40 (arenas < 1 TB so offsets fit 40 bits; literal runs
// < 2^24); b = dst.
struct DevLit { uint64_t a, b; };
// resolved match whose pattern (<= 8 bytes) the host inlined from the
// compressed literal bytes: write-only, no scattered pattern reads.
// meta = dst | len << 40 | period << 52 (len <= 256, period <= 8).
struct DevBrInl { uint64_t meta, pat; };
// contains window: a value-aligned run of consecutive non-null values of one
// PLAIN byte-array page, <= CWIN bytes total (or a single oversized value,
// nbytes > CWIN). Host builds these at load time by decompressing the page
// and walking the length chain once — the kernel has NO serial spine.
struct DevCWin {
  uint64_t src;       // absolute dec-arena offset of the first length field
  uint64_t starts;    // index into the u16 window-relative starts pool
  uint32_t nbytes;    // total bytes ([len][bytes] records) in this window
  uint32_t n_values;
  uint32_t dense0;    // non-null value index of value 0 within its page
  int32_t page_id;
};

// comparison kernel ops (matches gpuq_op order where applicable)
enum CmpMode { CMP_EQ = 0, CMP_NE, CMP_LT, CMP_LE, CMP_GT, CMP_GE, CMP_RANGE };

enum { AGGK_COUNT_STAR = 0, AGGK_COUNT, AGGK_SUM_I64, AGGK_SUM_F64,
       AGGK_MIN_I64, AGGK_MAX_I64, AGGK_MIN_F64, AGGK_MAX_F64,
       AGGK_MIN_RANK, AGGK_MAX_RANK,   // utf8 min/max via dict sort-ranks
       AGGK_MIN_STR, AGGK_MAX_STR };   // utf8 min/max via strrefs (hash cols)

constexpr int MAX_KEYS = 4;
constexpr int MAX_AGGS = 8;

// Aggregate table layout: G groups x (1 + 2*n_aggs) u64 slots:
//   slot 0: presence (selected-row count)
//   per agg a: slot 1+2a = value (i64 / f64 bits), slot 2+2a = non-null count
//
// f64 SUM is exact and order-independent: each value is decomposed into a
// 256-bit fixed-point two's-complement contribution (lsb weight 2^-160) and
// accumulated limb-wise with carry-propagating u64 atomics into `fsum`
// ([fsum_n][n_groups][4] limbs); the export rounds the exact 256-bit total
// to the nearest double once. Covers |x| in [2^-160, 2^86) — far beyond log
// analytics ranges; out-of-window values set *err = ERR_FSUM_RANGE (loud).
struct AggArgs {
  const uint8_t* mask;          // selection mask (1 byte/row), may be null (=all)
  int64_t n_rows;               // partition rows
  int n_keys;
  const int32_t* key_gid[MAX_KEYS];  // 0 = NULL group
  int32_t key_size[MAX_KEYS];        // global dict size incl. null slot
  int n_aggs;
  int32_t agg_kind[MAX_AGGS];        // AGGK_*
  const int64_t* agg_val[MAX_AGGS];  // i64 array or f64 bits (same width)
  const uint8_t* agg_valid[MAX_AGGS];// may be null (= all valid)
  uint8_t cnt_skip[MAX_AGGS];        // footer null_count==0 for every chunk:
                                     // count == presence, skip count atomics
  int32_t fsum_idx[MAX_AGGS];        // agg -> fsum table index (-1 = none)
  int32_t fsum_n;                    // number of f64-sum aggregates
  uint64_t* fsum;                    // superaccumulators (zeroed before launch)
  const uint8_t* dec;                // arena base (strref compares/hashes)
  uint64_t* table;
  int32_t n_groups;
  int32_t* err;
};

// error codes written to *d_error by kernels
enum { ERR_NONE = 0, ERR_LZ4 = 1, ERR_RLE = 2, ERR_DELTA = 3, ERR_DICT_RANGE = 4,
       ERR_PAGE = 5, ERR_FSUM_RANGE = 6, ERR_HASH_CAP = 7, ERR_HASH_PROBE = 8,
       ERR_PAIR_CAP = 9, ERR_PAIR_PROBE = 10 };

}  // namespace gpuq
