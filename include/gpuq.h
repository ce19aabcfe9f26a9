/* gpuq — MI355X-native query-execution path for Parseable log streams.
 *
 * C-ABI drop-in boundary for the DataFusion physical-plan execution used by
 * the reference's src/query. Each entry point states the reference interface
 * it replaces (file:line in parseablehq/parseable v2.9.5). A Rust
 * `ExecutionPlan` shim can wrap this unchanged (see INTEGRATION.md for the
 * binding stub a maintainer would add); today the C++/Python host harness in
 * parseable_amd/ drives it exactly where the reference's Rust would.
 *
 * Threading: all calls are thread-safe; gpuq_plan_execute may be called
 * concurrently for different partitions (mirrors DataFusion's
 * one-stream-per-partition polling, src/query/mod.rs:341-363).
 * Ownership: the library owns device memory; exported batches are host
 * Arrow buffers released via the ArrowArrayStream release callback
 * (Arrow C Stream Interface).
 */
#ifndef GPUQ_H
#define GPUQ_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct gpuq_ctx gpuq_ctx;
typedef struct gpuq_plan gpuq_plan;

/* ---- Arrow C Data/Stream Interface (stable ABI, restated per spec) ---- */
#ifndef ARROW_C_DATA_INTERFACE
#define ARROW_C_DATA_INTERFACE
struct ArrowSchema {
  const char* format;
  const char* name;
  const char* metadata;
  int64_t flags;
  int64_t n_children;
  struct ArrowSchema** children;
  struct ArrowSchema* dictionary;
  void (*release)(struct ArrowSchema*);
  void* private_data;
};
struct ArrowArray {
  int64_t length;
  int64_t null_count;
  int64_t offset;
  int64_t n_buffers;
  int64_t n_children;
  const void** buffers;
  struct ArrowArray** children;
  struct ArrowArray* dictionary;
  void (*release)(struct ArrowArray*);
  void* private_data;
};
#endif
#ifndef ARROW_C_STREAM_INTERFACE
#define ARROW_C_STREAM_INTERFACE
struct ArrowArrayStream {
  int (*get_schema)(struct ArrowArrayStream*, struct ArrowSchema* out);
  int (*get_next)(struct ArrowArrayStream*, struct ArrowArray* out);
  const char* (*get_last_error)(struct ArrowArrayStream*);
  void (*release)(struct ArrowArrayStream*);
  void* private_data;
};
#endif

/* ---- plan inputs (the predicate/aggregate IR of SURVEY.md §8b) ---- */

/* One parquet file of the stream, with the row groups selected by the
 * planner after manifest/min-max pruning — the product of
 * StandardTableProvider::scan's collect_from_snapshot + pruning
 * (stream_schema_provider.rs:505-600,1049-1137). row_groups==NULL selects
 * all row groups. */
typedef struct {
  const char* path;
  const int32_t* row_groups;
  int32_t n_row_groups;
} gpuq_file;

typedef enum {
  GPUQ_EQ = 0, GPUQ_NE, GPUQ_LT, GPUQ_LE, GPUQ_GT, GPUQ_GE,
  GPUQ_BETWEEN,       /* lo <= x <= hi (SQL BETWEEN); with hi_exclusive:
                         lo <= x < hi — the injected time filter shape of
                         query/mod.rs:829-888 */
  GPUQ_CONTAINS       /* LIKE '%lit%' byte substring */
} gpuq_op;

typedef enum { GPUQ_LIT_I64 = 0, GPUQ_LIT_F64, GPUQ_LIT_STR } gpuq_lit;

typedef struct {
  const char* column;
  int32_t op;          /* gpuq_op */
  int32_t lit_kind;    /* gpuq_lit */
  int64_t i64[2];      /* literal / BETWEEN bounds */
  double f64[2];
  const char* str;     /* utf8 literal (EQ.. / CONTAINS) */
  int32_t hi_exclusive;/* BETWEEN only */
} gpuq_pred;

typedef enum {
  GPUQ_AGG_COUNT_STAR = 0, GPUQ_AGG_COUNT, GPUQ_AGG_SUM,
  GPUQ_AGG_MIN, GPUQ_AGG_MAX
} gpuq_agg_op;

typedef struct { int32_t op; const char* column; } gpuq_agg;

/* ---- session -------------------------------------------------------- */

/* device_mask: bit i selects HIP device i. Replaces the per-query session
 * setup of Query::execute (query/mod.rs:152-165,291-372). */
gpuq_ctx* gpuq_session_create(uint64_t device_mask);
void gpuq_session_destroy(gpuq_ctx*);
const char* gpuq_last_error(gpuq_ctx*);

/* ---- plan ----------------------------------------------------------- */

/* Build the physical plan. Mirrors TableProvider::scan
 * (stream_schema_provider.rs:616-753): file list + projection + pushed-down
 * predicate conjunction + group-by/aggregate spec -> executable plan.
 * Schema is taken from the parquet footers (merged; must agree).
 * projection lists output columns for non-aggregate scans (may be NULL when
 * group_by/aggs are given); with no aggregates, a projection + limit runs a
 * TOP-K scan ordered by p_timestamp DESC (the console default; ordering
 * contract stream_schema_provider.rs:181-204) and the exported batch holds
 * the projected rows. limit < 0 means none. Returns NULL on error
 * (see gpuq_last_error). */
gpuq_plan* gpuq_plan_build(gpuq_ctx*,
    const gpuq_file* files, int32_t n_files,
    const char* const* projection, int32_t n_projection,
    const gpuq_pred* preds, int32_t n_preds,
    const char* const* group_by, int32_t n_group_by,
    const gpuq_agg* aggs, int32_t n_aggs,
    int64_t limit);

/* Plan straight from Parseable's catalog metadata:
 * parses <stream_dir>/stream.json (ObjectStoreFormat snapshot,
 * storage/mod.rs:335-380) and the daily manifest JSON
 * (catalog/manifest.rs:143-157), selects manifests by the time window
 * (Snapshot::manifests, catalog/snapshot.rs:42-71), prunes files by
 * per-column min/max TypedStatistics (can_be_pruned/satisfy_constraints,
 * stream_schema_provider.rs:1049-1137), and answers bare count(*) from
 * manifest num_rows sums (query.rs:189-256). The injected time range is
 * passed as a hi_exclusive BETWEEN on p_timestamp among `preds`.
 * *fast_count: >=0 manifest-answered count (no plan); -1 plan returned
 * (or error: check return + gpuq_last_error); -2 empty relation. */
gpuq_plan* gpuq_plan_build_from_stream(gpuq_ctx*, const char* stream_dir,
    const char* const* projection, int32_t n_projection,
    const gpuq_pred* preds, int32_t n_preds,
    const char* const* group_by, int32_t n_group_by,
    const gpuq_agg* aggs, int32_t n_aggs, int64_t limit,
    int64_t* fast_count);

/* Independent output partitions, one per selected device (the byte-balanced
 * row-group shards of balanced_file_groups, stream_schema_provider.rs:146-165,
 * collapsed to one stream per GPU). */
int32_t gpuq_plan_partition_count(gpuq_plan*);

/* Stage the partition's raw column chunks into device HBM (the hot-tier
 * residency step; excluded from the timed execute — PCIe-inclusive rates are
 * reported separately). Idempotent. */
int32_t gpuq_plan_load(gpuq_plan*, int32_t partition);

/* Execute one partition on its GPU: decompress -> decode -> filter ->
 * partial hash-aggregate, and export the PARTIAL aggregate table (or
 * projected rows) as one Arrow record batch through `out`. The caller
 * merges partials across partitions/nodes — the AggregateExec
 * Partial->Final split of the reference's engine.
 * Replaces ExecutionPlan::execute(partition, ctx) -> RecordBatch stream. */
int32_t gpuq_plan_execute(gpuq_plan*, int32_t partition,
                          struct ArrowArrayStream* out);

/* Plan metrics, summed over executed partitions. bytes_scanned = compressed
 * bytes of the column chunks read (the reference's ParquetExec
 * `bytes_scanned` plan metric, query/mod.rs:466-481);
 * rowgroup_bytes_total = total file bytes of scanned row groups (all
 * columns). kernel_ns = GPU time inside kernels (HIP events). */
typedef struct {
  int64_t rows_scanned;
  int64_t rows_out;
  int64_t bytes_scanned;
  int64_t rowgroup_bytes_total;
  int64_t hbm_bytes_est;      /* algorithmic HBM traffic estimate */
  int64_t kernel_ns;
  int64_t exec_ns;
  int64_t load_ns;
  int64_t decomp_ns;          /* decompression kernels only */
  int64_t cache_hit_bytes;    /* compressed bytes served from the GPU-resident
                                 hot tier (decompressed chunk images cached
                                 across plans/queries, SURVEY §8f-3;
                                 keyed like hottier.rs:1405-1417) */
} gpuq_metrics;
int32_t gpuq_plan_metrics(gpuq_plan*, gpuq_metrics* out);

void gpuq_plan_destroy(gpuq_plan*);

/* Library self-description: number of visible HIP devices (-1 on failure —
 * callers on GPU hosts must treat failure as fatal, never fall back). */
int32_t gpuq_device_count(void);

#ifdef __cplusplus
}
#endif
#endif /* GPUQ_H */
