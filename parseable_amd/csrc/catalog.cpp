#include "catalog.h"
#include "json.h"

#include <cstring>
#include <ctime>
#include <fstream>
#include <dlfcn.h>
#include <sstream>

namespace gpuq {
namespace {

// zstd frame decode via the system libzstd (the reference binds the same
// library through the zstd crate, catalog/manifest.rs ZSTD_LEVEL=3).
// dlopen'd lazily so plain-JSON-only deployments never need it.
std::string zstd_decompress_str(const std::string& in) {
  static void* h = dlopen("libzstd.so.1", RTLD_NOW);
  if (!h) throw std::runtime_error("libzstd.so.1 not available for compressed manifest");
  using FBound = unsigned long long (*)(const void*, size_t);
  using FDec = size_t (*)(void*, size_t, const void*, size_t);
  using FErr = unsigned (*)(size_t);
  static auto f_size = (FBound)dlsym(h, "ZSTD_getFrameContentSize");
  static auto f_dec = (FDec)dlsym(h, "ZSTD_decompress");
  static auto f_err = (FErr)dlsym(h, "ZSTD_isError");
  if (!f_size || !f_dec || !f_err)
    throw std::runtime_error("libzstd symbols missing");
  unsigned long long need = f_size(in.data(), in.size());
  size_t cap = (need + 1 < 2) ? (in.size() * 32 + (1 << 20))  // unknown size
                              : (size_t)need;
  for (;;) {
    std::string out(cap, '\0');
    size_t n = f_dec(out.data(), out.size(), in.data(), in.size());
    if (!f_err(n)) {
      out.resize(n);
      return out;
    }
    if (need + 1 >= 2 || cap > (1ull << 32))
      throw std::runtime_error("zstd manifest decompression failed");
    cap *= 4;  // content size unknown: grow and retry
  }
}

// Manifest bytes -> JSON text, sniffing the zstd magic exactly as
// decode_manifest does (catalog/manifest.rs:53-110: ZSTD_MAGIC
// 28 B5 2F FD; plain pre-compression manifests stay readable as-is).
std::string decode_manifest_text(std::string raw) {
  if (raw.size() >= 4 && (uint8_t)raw[0] == 0x28 && (uint8_t)raw[1] == 0xB5 &&
      (uint8_t)raw[2] == 0x2F && (uint8_t)raw[3] == 0xFD)
    return zstd_decompress_str(raw);
  return raw;
}

std::string read_file(const std::string& path) {
  std::ifstream in(path, std::ios::binary);
  if (!in) throw std::runtime_error("cannot open " + path);
  std::ostringstream ss;
  ss << in.rdbuf();
  return ss.str();
}

// chrono serde format: 2025-09-01T00:00:00.000000Z -> ms since epoch
int64_t parse_iso_ms(const std::string& s) {
  struct tm tm {};
  double frac = 0;
  // YYYY-MM-DDTHH:MM:SS[.ffffff]Z
  if (s.size() < 19) throw std::runtime_error("bad timestamp: " + s);
  tm.tm_year = std::stoi(s.substr(0, 4)) - 1900;
  tm.tm_mon = std::stoi(s.substr(5, 2)) - 1;
  tm.tm_mday = std::stoi(s.substr(8, 2));
  tm.tm_hour = std::stoi(s.substr(11, 2));
  tm.tm_min = std::stoi(s.substr(14, 2));
  tm.tm_sec = std::stoi(s.substr(17, 2));
  size_t dot = s.find('.', 19 - 1);
  if (dot != std::string::npos) {
    size_t end = s.find_first_not_of("0123456789", dot + 1);
    frac = std::stod("0." + s.substr(dot + 1, end - dot - 1));
  }
  int64_t secs = timegm(&tm);
  return secs * 1000 + (int64_t)(frac * 1000.0 + 0.5);
}

struct Bound {
  bool has = false;
  int64_t lo = 0, hi = 0;  // [lo, hi) after normalization
};

// derive the ts window from predicates on p_timestamp (the injected
// `>= lo AND < hi` arrives as a hi-exclusive BETWEEN; explicit BETWEEN is
// inclusive -> hi+1)
Bound ts_window(const gpuq_pred* preds, int32_t n) {
  Bound b;
  for (int32_t i = 0; i < n; i++) {
    const auto& p = preds[i];
    if (!p.column || strcmp(p.column, "p_timestamp") != 0) continue;
    if (p.op == GPUQ_BETWEEN) {
      int64_t lo = p.i64[0];
      int64_t hi = p.hi_exclusive ? p.i64[1] : p.i64[1] + 1;
      if (!b.has) { b = {true, lo, hi}; }
      else { b.lo = std::max(b.lo, lo); b.hi = std::min(b.hi, hi); }
    }
  }
  return b;
}

// satisfy_constraints port (stream_schema_provider.rs:1111-1137): can any
// value in [min,max] match `op value`? can_be_pruned = !satisfy.
bool stats_can_match(const JValue& stats, const gpuq_pred& p, int64_t lit_i,
                     double lit_f, const char* lit_s, int op) {
  // externally tagged serde enum: {"Int": {...}} | {"Float"} | {"String"} | {"Bool"}
  if (stats.has("Int")) {
    if (p.lit_kind != GPUQ_LIT_I64) return true;  // type mismatch: cannot prune
    int64_t mn = stats.at("Int").at("min").as_i64();
    int64_t mx = stats.at("Int").at("max").as_i64();
    switch (op) {
      case GPUQ_EQ: return lit_i >= mn && lit_i <= mx;
      case GPUQ_LT: return mn < lit_i;
      case GPUQ_LE: return mn <= lit_i;
      case GPUQ_GT: return mx > lit_i;
      case GPUQ_GE: return mx >= lit_i;
    }
    return true;
  }
  if (stats.has("Float")) {
    if (p.lit_kind != GPUQ_LIT_F64) return true;
    double mn = stats.at("Float").at("min").as_f64();
    double mx = stats.at("Float").at("max").as_f64();
    switch (op) {
      case GPUQ_EQ: return lit_f >= mn && lit_f <= mx;
      case GPUQ_LT: return mn < lit_f;
      case GPUQ_LE: return mn <= lit_f;
      case GPUQ_GT: return mx > lit_f;
      case GPUQ_GE: return mx >= lit_f;
    }
    return true;
  }
  if (stats.has("String")) {
    if (p.lit_kind != GPUQ_LIT_STR || !lit_s) return true;
    const std::string& mn = stats.at("String").at("min").s;
    const std::string& mx = stats.at("String").at("max").s;
    std::string v(lit_s);
    switch (op) {
      case GPUQ_EQ: return v >= mn && v <= mx;
      case GPUQ_LT: return mn < v;
      case GPUQ_LE: return mn <= v;
      case GPUQ_GT: return mx > v;
      case GPUQ_GE: return mx >= v;
    }
    return true;
  }
  return true;  // Bool / unknown: never prune
}

// ManifestExt::can_be_pruned port (stream_schema_provider.rs:1049-1078):
// a file is pruned when some predicate provably matches no row. NE and
// CONTAINS never prune (the reference behaves the same); BETWEEN
// decomposes into its two bounds.
bool file_pruned(const JValue& fe, const gpuq_pred* preds, int32_t n) {
  const JValue* cols = fe.get("columns");
  if (!cols || cols->kind != JValue::ARR) return false;
  auto find_stats = [&](const char* name) -> const JValue* {
    for (const auto& c : cols->arr) {
      const JValue* nm = c->get("name");
      if (nm && nm->s == name) return c->get("stats");
    }
    return nullptr;
  };
  for (int32_t i = 0; i < n; i++) {
    const auto& p = preds[i];
    if (p.op == GPUQ_NE || p.op == GPUQ_CONTAINS || !p.column) continue;
    const JValue* st = find_stats(p.column);
    if (!st || st->kind != JValue::OBJ) continue;
    if (p.op == GPUQ_BETWEEN) {
      gpuq_pred lo = p; lo.op = GPUQ_GE;  // x >= lo
      if (!stats_can_match(*st, lo, p.i64[0], p.f64[0], nullptr, GPUQ_GE)) return true;
      int ophi = p.hi_exclusive ? GPUQ_LT : GPUQ_LE;
      if (!stats_can_match(*st, p, p.i64[1], p.f64[1], nullptr, ophi)) return true;
    } else {
      if (!stats_can_match(*st, p, p.i64[0], p.f64[0], p.str, p.op)) return true;
    }
  }
  return false;
}

}  // namespace

CatalogPlanInput catalog_plan(const std::string& stream_dir,
                              const gpuq_pred* preds, int32_t n_preds,
                              bool bare_count_star) {
  CatalogPlanInput out;
  std::string root = stream_dir;
  size_t slash = root.find_last_of('/');
  root = (slash == std::string::npos) ? "." : root.substr(0, slash);

  JPtr snap_doc = JsonParser(read_file(stream_dir + "/stream.json")).parse();
  const JValue& snapshot = snap_doc->at("snapshot");
  const JValue& mlist = snapshot.at("manifest_list");
  Bound win = ts_window(preds, n_preds);

  // value predicates excluding the window predicate itself (it is still a
  // pred for file-level ts pruning, so keep the full list for file_pruned)
  bool has_value_preds = false;
  for (int32_t i = 0; i < n_preds; i++)
    if (!preds[i].column || strcmp(preds[i].column, "p_timestamp") != 0)
      has_value_preds = true;

  std::vector<JPtr> kept_files;
  for (const auto& item : mlist.arr) {
    // Snapshot::manifests (snapshot.rs:42-71): keep manifests overlapping
    // the window
    if (win.has) {
      int64_t ub = parse_iso_ms(item->at("time_upper_bound").s);
      int64_t lb = parse_iso_ms(item->at("time_lower_bound").s);
      if (!(ub >= win.lo && lb < win.hi)) continue;
    }
    std::string mpath = item->at("manifest_path").s;
    if (!mpath.empty() && mpath[0] != '/') mpath = root + "/" + mpath;
    JPtr man = JsonParser(decode_manifest_text(read_file(mpath))).parse();
    for (const auto& fe : man->at("files").arr) {
      if (file_pruned(*fe, preds, n_preds)) continue;
      kept_files.push_back(fe);
    }
  }

  // count fast path: bare count(*), no value preds, and the window (if any)
  // fully covers every kept file's ts bounds
  if (bare_count_star && !has_value_preds) {
    bool exact = true;
    int64_t total = 0;
    for (const auto& fe : kept_files) {
      total += fe->at("num_rows").as_i64();
      if (!win.has) continue;
      const JValue* cols = fe->get("columns");
      const JValue* st = nullptr;
      if (cols)
        for (const auto& c : cols->arr) {
          const JValue* nm = c->get("name");
          if (nm && nm->s == "p_timestamp") { st = c->get("stats"); break; }
        }
      if (!st || !st->has("Int")) { exact = false; break; }
      int64_t mn = st->at("Int").at("min").as_i64();
      int64_t mx = st->at("Int").at("max").as_i64();
      if (!(win.lo <= mn && mx < win.hi)) { exact = false; break; }
    }
    if (exact) {
      out.fast_count = total;
      return out;
    }
  }

  for (const auto& fe : kept_files) {
    std::string fp = fe->at("file_path").s;
    out.files.push_back(fp.empty() || fp[0] == '/' ? fp : root + "/" + fp);
  }
  return out;
}

}  // namespace gpuq
