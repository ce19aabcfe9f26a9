#include "meta.h"
#include <cstring>

namespace gpuq {

// ---- thrift compact protocol ----------------------------------------
namespace {

struct Reader {
  const uint8_t *p, *end;
  uint8_t u8() {
    if (p >= end) throw std::runtime_error("thrift: eof");
    return *p++;
  }
  uint64_t varint() {
    uint64_t v = 0; int sh = 0;
    for (;;) {
      uint8_t b = u8();
      v |= (uint64_t)(b & 0x7f) << sh;
      if (!(b & 0x80)) return v;
      sh += 7;
    }
  }
  int64_t zigzag() { uint64_t v = varint(); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }
  // returns wire type, 0 = stop
  int field(int16_t& fid) {
    uint8_t b = u8();
    if (!b) return 0;
    int delta = b >> 4, t = b & 0xf;
    if (delta) fid = (int16_t)(fid + delta); else fid = (int16_t)zigzag();
    return t;
  }
  void list_head(int& etype, uint32_t& n) {
    uint8_t h = u8(); etype = h & 0xf; n = h >> 4;
    if (n == 15) n = (uint32_t)varint();
  }
  std::string binary() {
    uint64_t n = varint();
    if (p + n > end) throw std::runtime_error("thrift: binary overrun");
    std::string s((const char*)p, n); p += n; return s;
  }
  const uint8_t* binary_view(uint32_t& n) {
    n = (uint32_t)varint();
    if (p + n > end) throw std::runtime_error("thrift: binary overrun");
    const uint8_t* s = p; p += n; return s;
  }
  void skip(int t) {
    switch (t) {
      case 1: case 2: break;
      case 3: u8(); break;
      case 4: case 5: case 6: zigzag(); break;
      case 7: p += 8; break;
      case 8: { uint64_t n = varint(); p += n; break; }
      case 9: case 10: {
        int et; uint32_t n; list_head(et, n);
        for (uint32_t i = 0; i < n; i++) skip(et);
        break;
      }
      case 12: {
        int16_t fid = 0;
        for (;;) { int ft = field(fid); if (!ft) break; skip(ft); }
        break;
      }
      default: throw std::runtime_error("thrift: bad wire type");
    }
    if (p > end) throw std::runtime_error("thrift: overrun");
  }
};

void parse_stats(Reader& r, ColumnChunkMeta& cm, int phys_type) {
  // parquet Statistics struct: 5=max_value, 6=min_value (new);
  // 1=max, 2=min (deprecated) — all binary, little-endian for numerics;
  // 3=null_count (i64).
  int16_t fid = 0;
  std::string mn, mx;
  for (;;) {
    int t = r.field(fid);
    if (!t) break;
    if ((fid == 5 || fid == 1) && t == 8) mx = r.binary();
    else if ((fid == 6 || fid == 2) && t == 8) mn = r.binary();
    else if (fid == 3 && (t == 5 || t == 6)) cm.null_count = r.zigzag();
    else r.skip(t);
  }
  if ((phys_type == PT_INT64) && mn.size() == 8 && mx.size() == 8) {
    memcpy(&cm.stat_min, mn.data(), 8);
    memcpy(&cm.stat_max, mx.data(), 8);
    cm.has_i64_stats = true;
  }
}

void parse_column_meta(Reader& r, ColumnChunkMeta& cm, const FileMeta& fm, int& phys) {
  int16_t fid = 0;
  std::string name;
  for (;;) {
    int t = r.field(fid);
    if (!t) break;
    switch (fid) {
      case 1: phys = (int)r.zigzag(); break;
      case 3: {  // path_in_schema (flat: one element)
        int et; uint32_t n; r.list_head(et, n);
        for (uint32_t i = 0; i < n; i++) {
          std::string s = r.binary();
          if (i == 0) name = s;
        }
        break;
      }
      case 4: cm.codec = (int)r.zigzag(); break;
      case 5: cm.num_values = r.zigzag(); break;
      case 7: cm.total_compressed_size = r.zigzag(); break;
      case 9: cm.data_page_offset = r.zigzag(); break;
      case 11: cm.dict_page_offset = r.zigzag(); break;
      case 12: parse_stats(r, cm, phys); break;
      default: r.skip(t);
    }
  }
  for (size_t i = 0; i < fm.columns.size(); i++)
    if (fm.columns[i].name == name) { cm.schema_idx = (int)i; break; }
}

}  // namespace

FileMeta parse_footer(const uint8_t* buf, size_t len) {
  if (len < 12 || memcmp(buf + len - 4, "PAR1", 4))
    throw std::runtime_error("not a parquet file");
  uint32_t flen;
  memcpy(&flen, buf + len - 8, 4);
  if ((size_t)flen + 8 > len) throw std::runtime_error("bad footer length");
  Reader r{buf + len - 8 - flen, buf + len - 8};
  FileMeta fm;
  int16_t fid = 0;
  for (;;) {
    int t = r.field(fid);
    if (!t) break;
    switch (fid) {
      case 2: {  // schema: list<SchemaElement>
        int et; uint32_t n; r.list_head(et, n);
        for (uint32_t i = 0; i < n; i++) {
          int16_t f2 = 0;
          SchemaColumn sc;
          int rep = 0;
          for (;;) {
            int t2 = r.field(f2);
            if (!t2) break;
            switch (f2) {
              case 1: sc.phys_type = (int)r.zigzag(); break;
              case 3: rep = (int)r.zigzag(); break;
              case 4: sc.name = r.binary(); break;
              default: r.skip(t2);
            }
          }
          sc.optional = (rep == 1);
          if (i > 0) fm.columns.push_back(sc);  // element 0 = root group
        }
        break;
      }
      case 3: fm.num_rows = r.zigzag(); break;
      case 4: {  // row_groups
        int et; uint32_t n; r.list_head(et, n);
        for (uint32_t g = 0; g < n; g++) {
          RowGroupMeta rg;
          rg.chunks.resize(fm.columns.size());
          int16_t f2 = 0;
          for (;;) {
            int t2 = r.field(f2);
            if (!t2) break;
            if (f2 == 1) {  // columns: list<ColumnChunk>
              int et2; uint32_t nc; r.list_head(et2, nc);
              for (uint32_t c = 0; c < nc; c++) {
                ColumnChunkMeta cm;
                int phys = -1;
                int16_t f3 = 0;
                for (;;) {
                  int t3 = r.field(f3);
                  if (!t3) break;
                  if (f3 == 3) parse_column_meta(r, cm, fm, phys);
                  else r.skip(t3);
                }
                if (cm.schema_idx >= 0 && cm.schema_idx < (int)fm.columns.size()) {
                  rg.chunks[cm.schema_idx] = cm;
                  rg.total_compressed_size += cm.total_compressed_size;
                }
              }
            } else if (f2 == 2) rg.total_byte_size = r.zigzag();
            else if (f2 == 3) rg.num_rows = r.zigzag();
            else r.skip(t2);
          }
          fm.row_groups.push_back(std::move(rg));
        }
        break;
      }
      default: r.skip(t);
    }
  }
  return fm;
}

std::vector<PageInfo> walk_pages(const uint8_t* buf, const ColumnChunkMeta& cm,
                                 int64_t rg_rows) {
  std::vector<PageInfo> out;
  const uint8_t* p = buf + cm.start_offset();
  const uint8_t* chunk_end = buf + cm.start_offset() + cm.total_compressed_size;
  int64_t rows = 0;
  while (rows < rg_rows && p < chunk_end) {
    Reader r{p, chunk_end};
    PageInfo pi{};
    int16_t fid = 0;
    for (;;) {
      int t = r.field(fid);
      if (!t) break;
      switch (fid) {
        case 1: pi.type = (int)r.zigzag(); break;
        case 2: pi.uncomp_size = (int32_t)r.zigzag(); break;
        case 3: pi.comp_size = (int32_t)r.zigzag(); break;
        case 5: {  // DataPageHeader
          int16_t f2 = 0;
          for (;;) {
            int t2 = r.field(f2);
            if (!t2) break;
            if (f2 == 1) pi.num_values = (int32_t)r.zigzag();
            else if (f2 == 2) pi.encoding = (int)r.zigzag();
            else r.skip(t2);
          }
          break;
        }
        case 7: {  // DictionaryPageHeader
          int16_t f2 = 0;
          for (;;) {
            int t2 = r.field(f2);
            if (!t2) break;
            if (f2 == 1) pi.num_values = (int32_t)r.zigzag();
            else if (f2 == 2) pi.encoding = (int)r.zigzag();
            else r.skip(t2);
          }
          break;
        }
        case 8:
          throw std::runtime_error("data page v2 not produced by this writer");
        default: r.skip(t);
      }
    }
    pi.payload_off = r.p - buf;
    out.push_back(pi);
    p = r.p + pi.comp_size;
    if (pi.type == PAGE_DATA) rows += pi.num_values;
  }
  if (rows != rg_rows)
    throw std::runtime_error("page walk row-count mismatch");
  return out;
}

int lz4_decompress_host(const uint8_t* src, size_t src_len,
                        uint8_t* dst, size_t dst_cap) {
  const uint8_t *sp = src, *send = src + src_len;
  uint8_t *dp = dst, *dend = dst + dst_cap;
  while (sp < send) {
    uint8_t token = *sp++;
    size_t lit = token >> 4;
    if (lit == 15) {
      uint8_t b;
      do { if (sp >= send) return -1; b = *sp++; lit += b; } while (b == 255);
    }
    if (sp + lit > send || dp + lit > dend) return -1;
    memcpy(dp, sp, lit); sp += lit; dp += lit;
    if (sp >= send) break;  // last sequence carries only literals
    if (sp + 2 > send) return -1;
    size_t off = sp[0] | ((size_t)sp[1] << 8); sp += 2;
    if (off == 0 || (size_t)(dp - dst) < off) return -1;
    size_t mlen = token & 0xf;
    if (mlen == 15) {
      uint8_t b;
      do { if (sp >= send) return -1; b = *sp++; mlen += b; } while (b == 255);
    }
    mlen += 4;
    if (dp + mlen > dend) return -1;
    const uint8_t* mp = dp - off;
    for (size_t i = 0; i < mlen; i++) dp[i] = mp[i];  // overlap repeats
    dp += mlen;
  }
  return (int)(dp - dst);
}

// ---- snappy (parquet codec 1) ----------------------------------------
// Raw snappy block format: uvarint uncompressed length, then tagged
// elements (literals; copies with 1/2/4-byte offsets, overlap = pattern
// repeat exactly like LZ4). The reference's test deployments write snappy
// (docker-compose-test.yaml:45; the codec is a knob, cli.rs:484-491).

int snappy_decompress_host(const uint8_t* src, size_t comp, uint8_t* dst,
                           size_t dst_cap) {
  size_t s = 0, d = 0;
  uint64_t ulen = 0;
  int sh = 0;
  for (;;) {
    if (s >= comp) return -1;
    uint8_t b = src[s++];
    ulen |= (uint64_t)(b & 0x7f) << sh;
    if (!(b & 0x80)) break;
    sh += 7;
  }
  if (ulen > dst_cap) return -1;
  while (s < comp) {
    uint8_t tag = src[s++];
    int type = tag & 3;
    if (type == 0) {  // literal
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int extra = (int)len - 60;
        if (s + extra > comp) return -1;
        len = 0;
        for (int i = 0; i < extra; i++) len |= (size_t)src[s + i] << (8 * i);
        len += 1;
        s += extra;
      }
      if (s + len > comp || d + len > dst_cap) return -1;
      memcpy(dst + d, src + s, len);
      s += len;
      d += len;
    } else {
      size_t ml, off;
      if (type == 1) {
        ml = ((tag >> 2) & 7) + 4;
        if (s >= comp) return -1;
        off = ((size_t)(tag >> 5) << 8) | src[s++];
      } else if (type == 2) {
        ml = (tag >> 2) + 1;
        if (s + 2 > comp) return -1;
        off = src[s] | ((size_t)src[s + 1] << 8);
        s += 2;
      } else {
        ml = (tag >> 2) + 1;
        if (s + 4 > comp) return -1;
        off = src[s] | ((size_t)src[s + 1] << 8) | ((size_t)src[s + 2] << 16) |
              ((size_t)src[s + 3] << 24);
        s += 4;
      }
      if (off == 0 || off > d || d + ml > dst_cap) return -1;
      const uint8_t* mp = dst + d - off;
      for (size_t i = 0; i < ml; i++) dst[d + i] = mp[i];  // overlap repeats
      d += ml;
    }
  }
  return (int)d;
}

// Snappy pages ride the litpar machinery unchanged: every literal becomes a
// raw->dec copy record, every copy a host-resolved match (same interval-map
// composition as lz4_walk's litpar mode) — the GPU never parses snappy at
// all. fallback=true on piece explosion (caller host-decompresses the page
// and stages the image directly).
Lz4Plan snappy_walk(const uint8_t* src, size_t comp, size_t uncomp) {
  Lz4Plan plan;
  plan.litpar = true;
  size_t s = 0, d = 0;

  struct MapEntry { uint32_t start, len, off, piece_start, piece_n; };
  std::vector<MapEntry> M;
  constexpr int MAX_PIECES_PER_RECORD = 64;
  auto entry_at = [&](uint32_t x) -> size_t {
    size_t lo = 0, hi = M.size();
    while (lo < hi) {
      size_t mid = (lo + hi) / 2;
      if (M[mid].start + M[mid].len > x) hi = mid;
      else lo = mid + 1;
    }
    return lo;
  };
  auto resolve = [&](uint32_t a, uint32_t b, std::vector<Lz4Piece>& out) -> bool {
    size_t i = entry_at(a);
    uint32_t x = a;
    while (x < b) {
      if ((int)out.size() > MAX_PIECES_PER_RECORD) return false;
      if (i >= M.size() || M[i].start >= b) {
        out.push_back({x, b - x});
        break;
      }
      const MapEntry& e = M[i];
      if (x < e.start) {
        out.push_back({x, e.start - x});
        x = e.start;
        continue;
      }
      uint32_t y = std::min<uint32_t>(b, e.start + e.len);
      while (x < y) {
        if ((int)out.size() > MAX_PIECES_PER_RECORD) return false;
        uint32_t p = (x - e.start) % e.off;
        uint32_t chunk = std::min(y - x, e.off - p);
        uint32_t po = 0, q = p, left = chunk;
        for (uint32_t k = 0; k < e.piece_n && left; k++) {
          const Lz4Piece& pc = plan.pieces[e.piece_start + k];
          if (q < po + pc.len) {
            uint32_t within = q - po;
            uint32_t take = std::min(pc.len - within, left);
            out.push_back({pc.src + within, take});
            q += take;
            left -= take;
          }
          po += pc.len;
        }
        if (left) return false;
        x += chunk;
      }
      i++;
    }
    return true;
  };
  auto defer_match = [&](size_t dst, size_t off, size_t ml) {
    uint32_t pat = (uint32_t)std::min(off, ml);
    std::vector<Lz4Piece> pieces;
    if (resolve((uint32_t)(dst - off), (uint32_t)(dst - off + pat), pieces)) {
      uint32_t ps = (uint32_t)plan.pieces.size();
      plan.pieces.insert(plan.pieces.end(), pieces.begin(), pieces.end());
      plan.resolved.push_back({(uint32_t)dst, (uint32_t)ml, (uint32_t)off,
                               ps, (uint32_t)pieces.size()});
      M.push_back({(uint32_t)dst, (uint32_t)ml, (uint32_t)off, ps,
                   (uint32_t)pieces.size()});
    } else {
      plan.fallback = true;
    }
  };

  uint64_t ulen = 0;
  int sh = 0;
  for (;;) {
    if (s >= comp) throw std::runtime_error("snappy walk: eof in preamble");
    uint8_t b = src[s++];
    ulen |= (uint64_t)(b & 0x7f) << sh;
    if (!(b & 0x80)) break;
    sh += 7;
  }
  if (ulen != uncomp) throw std::runtime_error("snappy walk: size mismatch");
  while (s < comp) {
    plan.n_seq++;
    uint8_t tag = src[s++];
    int type = tag & 3;
    if (type == 0) {
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int extra = (int)len - 60;
        if (s + extra > comp) throw std::runtime_error("snappy walk: eof");
        len = 0;
        for (int i = 0; i < extra; i++) len |= (size_t)src[s + i] << (8 * i);
        len += 1;
        s += extra;
      }
      if (s + len > comp || d + len > uncomp)
        throw std::runtime_error("snappy walk: overrun");
      plan.lits.push_back({(uint32_t)d, (uint32_t)s, (uint32_t)len});
      s += len;
      d += len;
    } else {
      size_t ml, off;
      if (type == 1) {
        ml = ((tag >> 2) & 7) + 4;
        if (s >= comp) throw std::runtime_error("snappy walk: eof");
        off = ((size_t)(tag >> 5) << 8) | src[s++];
      } else if (type == 2) {
        ml = (tag >> 2) + 1;
        if (s + 2 > comp) throw std::runtime_error("snappy walk: eof");
        off = src[s] | ((size_t)src[s + 1] << 8);
        s += 2;
      } else {
        ml = (tag >> 2) + 1;
        if (s + 4 > comp) throw std::runtime_error("snappy walk: eof");
        off = src[s] | ((size_t)src[s + 1] << 8) | ((size_t)src[s + 2] << 16) |
              ((size_t)src[s + 3] << 24);
        s += 4;
      }
      if (off == 0 || off > d || d + ml > uncomp)
        throw std::runtime_error("snappy walk: bad copy");
      defer_match(d, off, ml);
      if (plan.fallback) return plan;
      d += ml;
    }
  }
  if (d != uncomp) throw std::runtime_error("snappy walk: size mismatch");
  return plan;
}

Lz4Plan lz4_walk(const uint8_t* src, size_t comp, size_t uncomp,
                 uint32_t seg_max, bool litpar) {
  Lz4Plan plan;
  plan.litpar = litpar;
  size_t s = 0, d = 0;
  uint32_t seg_s = 0, seg_d = 0;

  // interval map of deferred (gap) regions, sorted and disjoint. Each entry
  // mirrors a resolved record: bytes [start, start+len) have value
  // pattern[(x-start) % off], pattern described by literal pieces.
  struct MapEntry { uint32_t start, len, off, piece_start, piece_n; };
  std::vector<MapEntry> M;
  constexpr int MAX_PIECES_PER_RECORD = 64;

  auto entry_at = [&](uint32_t x) -> size_t {
    // first entry with start+len > x
    size_t lo = 0, hi = M.size();
    while (lo < hi) {
      size_t mid = (lo + hi) / 2;
      if (M[mid].start + M[mid].len > x) hi = mid;
      else lo = mid + 1;
    }
    return lo;
  };
  auto overlaps_gap = [&](uint32_t a, uint32_t b) {
    size_t i = entry_at(a);
    return i < M.size() && M[i].start < b;
  };
  // decompose [a, b) into literal-source pieces; false on piece explosion
  auto resolve = [&](uint32_t a, uint32_t b, std::vector<Lz4Piece>& out) -> bool {
    size_t i = entry_at(a);
    uint32_t x = a;
    while (x < b) {
      if ((int)out.size() > MAX_PIECES_PER_RECORD) return false;
      if (i >= M.size() || M[i].start >= b) {
        out.push_back({x, b - x});           // literal region
        x = b;
        break;
      }
      const MapEntry& e = M[i];
      if (x < e.start) {
        out.push_back({x, e.start - x});     // literal prefix
        x = e.start;
        continue;
      }
      uint32_t y = std::min<uint32_t>(b, e.start + e.len);
      while (x < y) {
        if ((int)out.size() > MAX_PIECES_PER_RECORD) return false;
        uint32_t p = (x - e.start) % e.off;  // pattern position
        uint32_t chunk = std::min(y - x, e.off - p);
        // map [p, p+chunk) through e's pieces (pattern offsets = prefix sums)
        uint32_t po = 0, q = p, left = chunk;
        for (uint32_t k = 0; k < e.piece_n && left; k++) {
          const Lz4Piece& pc = plan.pieces[e.piece_start + k];
          if (q < po + pc.len) {
            uint32_t within = q - po;
            uint32_t take = std::min(pc.len - within, left);
            out.push_back({pc.src + within, take});
            q += take;
            left -= take;
          }
          po += pc.len;
        }
        if (left) return false;              // pattern pieces inconsistent
        x += chunk;
      }
      i++;
    }
    return true;
  };
  auto defer_match = [&](size_t dst, size_t off, size_t ml) {
    plan.backrefs.push_back({(uint32_t)dst, (uint32_t)(dst - off), (uint32_t)ml});
    if (!plan.fallback) {
      uint32_t pat = (uint32_t)std::min(off, ml);
      std::vector<Lz4Piece> pieces;
      if (resolve((uint32_t)(dst - off), (uint32_t)(dst - off + pat), pieces)) {
        uint32_t ps = (uint32_t)plan.pieces.size();
        plan.pieces.insert(plan.pieces.end(), pieces.begin(), pieces.end());
        plan.resolved.push_back({(uint32_t)dst, (uint32_t)ml, (uint32_t)off,
                                 ps, (uint32_t)pieces.size()});
        // insert map entry (records arrive in ascending dst order)
        M.push_back({(uint32_t)dst, (uint32_t)ml, (uint32_t)off, ps,
                     (uint32_t)pieces.size()});
      } else {
        plan.fallback = true;
        plan.resolved.clear();
        plan.pieces.clear();
      }
    }
  };

  auto close_segment = [&](size_t end_s, size_t end_d, bool big) {
    if (end_d > seg_d || end_s > seg_s) {
      plan.segs.push_back({seg_s, seg_d, (uint32_t)(end_s - seg_s),
                           (uint32_t)(end_d - seg_d), (uint8_t)big});
    }
    seg_s = (uint32_t)end_s;
    seg_d = (uint32_t)end_d;
  };
  while (s < comp && d < uncomp) {
    size_t seq_s = s, seq_d = d;
    plan.n_seq++;
    uint8_t token = src[s++];
    size_t lit = token >> 4;
    if (lit == 15) {
      uint8_t b;
      do { if (s >= comp) throw std::runtime_error("lz4 walk: eof"); b = src[s++]; lit += b; } while (b == 255);
    }
    if (s + lit > comp || d + lit > uncomp) throw std::runtime_error("lz4 walk: overrun");
    size_t lit_src = s;
    size_t ml = 0, off = 0;
    bool has_match = false;
    size_t after_lit_s = s + lit;
    if (after_lit_s < comp) {
      if (after_lit_s + 2 > comp) throw std::runtime_error("lz4 walk: bad tail");
      off = src[after_lit_s] | ((size_t)src[after_lit_s + 1] << 8);
      size_t p2 = after_lit_s + 2;
      ml = token & 0xf;
      if (ml == 15) {
        uint8_t b;
        do { if (p2 >= comp) throw std::runtime_error("lz4 walk: eof"); b = src[p2++]; ml += b; } while (b == 255);
      }
      ml += 4;
      has_match = true;
      if (off == 0 || off > d + lit) throw std::runtime_error("lz4 walk: bad offset");
      s = p2;
    } else {
      s = after_lit_s;
    }
    size_t seq_out = lit + ml;
    if (d + seq_out > uncomp && has_match) throw std::runtime_error("lz4 walk: output overrun");

    if (litpar) {
      if (lit) plan.lits.push_back({(uint32_t)d, (uint32_t)lit_src, (uint32_t)lit});
      d += lit;
      if (has_match && ml) {
        defer_match(d, off, ml);
        if (plan.fallback) return plan;  // caller keeps the segment plan
        d += ml;
      }
      continue;
    }
    if (seq_out > seg_max) {
      close_segment(seq_s, seq_d, false);
      if (has_match && ml) defer_match(d + lit, off, ml);
      d += seq_out;
      close_segment(s, d, true);
      continue;
    }
    if ((d + seq_out) - seg_d > seg_max) {
      close_segment(seq_s, seq_d, false);
    }
    d += lit;
    if (has_match && ml) {
      size_t m_src = d - off;
      bool defer = m_src < seg_d ||
                   overlaps_gap((uint32_t)m_src,
                                (uint32_t)std::min(d, m_src + ml));
      if (defer) defer_match(d, off, ml);
      d += ml;
    }
  }
  if (d != uncomp) throw std::runtime_error("lz4 walk: size mismatch");
  if (!litpar) close_segment(s, d, false);
  return plan;
}

}  // namespace gpuq
