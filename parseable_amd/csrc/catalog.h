// Native catalog planner: parse Parseable's
// stream.json snapshot + daily manifest JSON, select manifests by the time
// predicates (Snapshot::manifests, src/catalog/snapshot.rs:42-71), prune
// files by per-column min/max TypedStatistics
// (ManifestExt::can_be_pruned / satisfy_constraints,
// src/query/stream_schema_provider.rs:1049-1137), and answer bare
// count(*) from manifest num_rows sums (the count fast path,
// src/handlers/http/query.rs:189-256, src/query/mod.rs:537-590).
#pragma once
#include "../../include/gpuq.h"
#include <cstdint>
#include <string>
#include <vector>

namespace gpuq {

struct CatalogPlanInput {
  std::vector<std::string> files;  // absolute paths of surviving files
  int64_t fast_count = -1;         // >= 0: answered without a scan
};

// preds: the full conjunction including the injected time range
// (a hi-exclusive BETWEEN on p_timestamp). Throws on malformed metadata.
CatalogPlanInput catalog_plan(const std::string& stream_dir,
                              const gpuq_pred* preds, int32_t n_preds,
                              bool bare_count_star);

}  // namespace gpuq
