// Host-side parquet metadata for the gpuq execution path: thrift-compact
// footer parse, page-header walk, and scalar LZ4_RAW (for the small
// dictionary pages the planner needs on the host — per-row work stays on
// the GPU). Implements the parquet-format spec as written by the
// reference's writer (src/parseable/streams.rs:705-780; parquet crate
// 58.1.0 pinned in the reference's Cargo.lock).
#pragma once
#include <cstdint>
#include <string>
#include <vector>
#include <stdexcept>

namespace gpuq {

enum PhysType { PT_BOOLEAN = 0, PT_INT32 = 1, PT_INT64 = 2, PT_INT96 = 3,
                PT_FLOAT = 4, PT_DOUBLE = 5, PT_BYTE_ARRAY = 6, PT_FLBA = 7 };
enum Encoding { ENC_PLAIN = 0, ENC_PLAIN_DICT = 2, ENC_RLE = 3,
                ENC_DELTA_BP = 5, ENC_RLE_DICT = 8 };
enum Codec { CODEC_UNCOMPRESSED = 0, CODEC_SNAPPY = 1, CODEC_LZ4_RAW = 7 };
enum PageType { PAGE_DATA = 0, PAGE_INDEX = 1, PAGE_DICT = 2, PAGE_DATA_V2 = 3 };

struct SchemaColumn {
  std::string name;
  int phys_type = -1;
  bool optional = false;   // OPTIONAL repetition -> max_def_level 1 (flat schema)
};

struct ColumnChunkMeta {
  int schema_idx = -1;
  int64_t data_page_offset = 0;
  int64_t dict_page_offset = -1;
  int64_t total_compressed_size = 0;
  int64_t num_values = 0;
  int codec = CODEC_UNCOMPRESSED;
  // footer statistics (row-group level min/max), for pruning and for
  // chunk-level predicate elision (all-rows-satisfy proofs need
  // null_count == 0; -1 = not present in the footer)
  bool has_i64_stats = false;
  int64_t stat_min = 0, stat_max = 0;
  int64_t null_count = -1;
  int64_t start_offset() const {
    return (dict_page_offset >= 0 && dict_page_offset < data_page_offset)
               ? dict_page_offset : data_page_offset;
  }
};

struct RowGroupMeta {
  int64_t num_rows = 0;
  int64_t total_byte_size = 0;              // uncompressed (footer field 2)
  int64_t total_compressed_size = 0;        // sum over chunks
  std::vector<ColumnChunkMeta> chunks;      // schema order
};

struct FileMeta {
  std::vector<SchemaColumn> columns;
  std::vector<RowGroupMeta> row_groups;
  int64_t num_rows = 0;
  int col_index(const std::string& name) const {
    for (size_t i = 0; i < columns.size(); i++)
      if (columns[i].name == name) return (int)i;
    return -1;
  }
};

struct PageInfo {
  int type;                 // PageType
  int64_t payload_off;      // absolute file offset of page payload
  int32_t comp_size, uncomp_size;
  int32_t num_values;
  int encoding;
};

// Parse footer thrift into FileMeta. Throws std::runtime_error.
FileMeta parse_footer(const uint8_t* buf, size_t len);

// Walk page headers of one column chunk (buf = whole file mapping).
std::vector<PageInfo> walk_pages(const uint8_t* buf, const ColumnChunkMeta& cm,
                                 int64_t rg_rows);

// Scalar LZ4 raw-block decode (host; dict pages + validation).
// Returns decompressed byte count or -1.
int lz4_decompress_host(const uint8_t* src, size_t src_len,
                        uint8_t* dst, size_t dst_cap);

// ---- segment plan for parallel GPU decompression -------------------
// The host walks the LZ4 sequence structure once (load-time metadata, the
// page-index analog) and splits each page into segments of <= seg_max
// output bytes at sequence boundaries. Matches whose source lies before
// the segment start — or overlaps a byte such a match should have written
// (transitive gaps) — become explicit backref records resolved by a second
// kernel in order. All parsing and byte movement still happens on the GPU
// every execution; the walk only yields offsets.
struct Lz4Segment {
  uint32_t s_off;      // into the page's compressed bytes
  uint32_t d_off;      // into the page's decompressed image
  uint32_t comp_len;
  uint32_t out_len;
  uint8_t big;         // single giant sequence: decompress straight to global
};
struct Lz4Backref {
  uint32_t dst, src;   // page-relative decompressed offsets (src < dst)
  uint32_t len;
};

// literal-resolved deferred match: out[dst + i] = pattern[i % off] where the
// pattern (min(off, len) bytes) is described by pieces whose sources are all
// PHASE-1 bytes (literals / in-segment matches) — so every record resolves
// in one parallel launch with no ordering. Built by composing the interval
// maps of earlier records during the host walk.
struct Lz4Resolved {
  uint32_t dst, len, off;
  uint32_t piece_start, piece_n;   // into Lz4Plan::pieces
};
struct Lz4Piece {
  uint32_t src, len;               // page-relative literal-region source
};
// litpar mode: pages whose content is a dense stream of short sequences
// (e.g. LZ4 over near-random dictionary indices) degenerate the segment
// kernel into a serial token parse. For those pages the walk instead emits
// one literal-copy record per sequence (compressed src -> decompressed dst,
// both page-relative) and defers EVERY match to the resolved-record pass —
// no segments, no serial parse on the GPU at all.
struct Lz4Lit {
  uint32_t dst, src, len;
};
struct Lz4Plan {
  std::vector<Lz4Segment> segs;
  std::vector<Lz4Backref> backrefs;    // fallback (serial window) records
  std::vector<Lz4Resolved> resolved;   // literal-resolved records
  std::vector<Lz4Piece> pieces;
  std::vector<Lz4Lit> lits;            // litpar mode only
  bool fallback = false;               // use `backrefs` for the whole page
  bool litpar = false;
  uint32_t n_seq = 0;                  // sequence count (mode decision)
};
// Throws on malformed streams. seg_max must match the kernel's LDS buffer.
// litpar=true produces the all-literal/all-resolved plan (fallback=true if
// a record's piece decomposition explodes — caller keeps the segment plan).
Lz4Plan lz4_walk(const uint8_t* src, size_t comp, size_t uncomp,
                 uint32_t seg_max, bool litpar = false);

// Snappy (parquet codec 1): scalar host decode, and the litpar-only walk
// whose records ride the SAME device kernels as LZ4's litpar mode.
int snappy_decompress_host(const uint8_t* src, size_t comp, uint8_t* dst,
                           size_t dst_cap);
Lz4Plan snappy_walk(const uint8_t* src, size_t comp, size_t uncomp);

}  // namespace gpuq
