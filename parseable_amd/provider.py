"""The drop-in host surface of the MI355X query path, mirroring the
reference's TableProvider/ExecutionPlan shape:

  StandardTableProvider.scan(query)  ->  GpuExecutionPlan
      mirrors StandardTableProvider::scan (stream_schema_provider.rs:616-753):
      snapshot -> manifest selection by time (snapshot.rs:42-71) -> file-level
      min/max pruning (can_be_pruned/satisfy_constraints,
      stream_schema_provider.rs:1049-1137) -> count fast path
      (query.rs:189-256, query/mod.rs:537-590) -> gpuq_plan_build.

  GpuExecutionPlan.execute(partition) -> pyarrow.RecordBatch (partial agg)
      mirrors ExecutionPlan::execute's pull-based per-partition streams
      (query/mod.rs:341-363); merge_partials() is the AggregateExec(Final).

  Query.execute() is the single execution funnel (query/mod.rs:152-165).

Time-range injection (`p_timestamp >= start AND < end`) mirrors `transform`
(query/mod.rs:829-888). All per-row compute runs in libgpuq.so on the GPU;
this layer only handles metadata (manifest JSON, footers are parsed in C++).
"""

from __future__ import annotations

import ctypes as C
import json
import os

from . import _lib
from ._lib import AGGS, OPS, GpuqAgg, GpuqFile, GpuqPred, GpuqMetrics


class GpuqError(RuntimeError):
    pass


class GpuSession:
    def __init__(self, device_mask: int = 0):
        import os

        lib = _lib.load()
        n = lib.gpuq_device_count()
        if n <= 0 and os.environ.get("GPUQ_FAKE_DEVICE"):
            n = 1  # host-plan debugging only; execute will fail at HIP calls
        if n <= 0:
            raise GpuqError(
                "no HIP devices visible — the gpuq path requires an MI355X; "
                "it never falls back to CPU"
            )
        self._lib = lib
        self._ctx = lib.gpuq_session_create(C.c_uint64(device_mask))
        if not self._ctx:
            raise GpuqError("gpuq_session_create failed")

    def _err(self):
        return self._lib.gpuq_last_error(self._ctx).decode()

    def close(self):
        if self._ctx:
            self._lib.gpuq_session_destroy(self._ctx)
            self._ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def _stats_entry(col):
    if not col or col.get("stats") is None:
        return None
    return col["stats"]


def _file_pruned(file_entry, preds):
    """Port of ManifestExt::can_be_pruned (stream_schema_provider.rs:1049-1078):
    a file is pruned when some predicate provably matches no row. BETWEEN is
    decomposed into ge(lo) + le/lt(hi) bounds as the reference's injected
    filters are (query/mod.rs:829-888)."""
    cols = {c["name"]: c for c in file_entry.get("columns", [])}

    def check(col, op, value):
        st = _stats_entry(cols.get(col))
        if st is None:
            return False  # no stats -> cannot prune
        (kind, mm), = st.items()
        mn, mx = mm["min"], mm["max"]
        if kind == "Int":
            if not isinstance(value, int) or isinstance(value, bool):
                return False
        elif kind == "Float":
            if not isinstance(value, float):
                return False
        elif kind == "String":
            if not isinstance(value, str):
                return False
        else:
            return False
        if op == "eq":
            sat = mn <= value <= mx
        elif op == "lt":
            sat = mn < value
        elif op == "le":
            sat = mn <= value
        elif op == "gt":
            sat = mx > value
        elif op == "ge":
            sat = mx >= value
        else:
            return False  # ne/contains: never pruned (reference does the same)
        return not sat

    for p in preds:
        op = p["op"]
        col = p["col"]
        if op == "between":
            if check(col, "ge", p["lo"]) or check(col, "le" if not p.get("hi_exclusive") else "lt", p["hi"]):
                return True
        elif op == "contains" or op == "ne":
            continue
        else:
            if check(col, op, p["lit"]):
                return True
    return False


ZSTD_MAGIC = b"\x28\xb5\x2f\xfd"


def _zstd_decompress(raw: bytes) -> bytes:
    """zstd frame decode via the system libzstd (the reference binds the
    same library through the zstd crate)."""
    import ctypes

    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_getFrameContentSize.restype = ctypes.c_ulonglong
    z.ZSTD_decompress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint
    need = z.ZSTD_getFrameContentSize(raw, len(raw))
    cap = len(raw) * 32 + (1 << 20) if need + 1 < 2 else int(need)
    while True:
        buf = ctypes.create_string_buffer(cap)
        n = z.ZSTD_decompress(buf, cap, raw, len(raw))
        if not z.ZSTD_isError(n):
            return buf.raw[:n]
        if need + 1 >= 2 or cap > 1 << 32:
            raise RuntimeError("zstd manifest decompression failed")
        cap *= 4


def _load_manifest(path):
    """Manifest bytes -> dict, sniffing the zstd magic exactly as the
    reference's decode_manifest (catalog/manifest.rs:53-110): compressed and
    plain JSON manifests share names; the leading frame magic is the only
    distinction."""
    with open(path, "rb") as fh:
        raw = fh.read()
    if raw[:4] == ZSTD_MAGIC:
        raw = _zstd_decompress(raw)
    return json.loads(raw)


def _parse_iso_ms(s):
    """chrono serde emits both '...%S.%fZ' and (when the fractional part is
    zero) '...%SZ' — accept both, matching catalog.cpp parse_iso_ms."""
    from datetime import datetime, timezone

    try:
        dt = datetime.strptime(s, "%Y-%m-%dT%H:%M:%S.%fZ")
    except ValueError:
        dt = datetime.strptime(s, "%Y-%m-%dT%H:%M:%SZ")
    return int(dt.replace(tzinfo=timezone.utc).timestamp() * 1000)


def supports_filters_pushdown(preds):
    """Mirror of TableProvider::supports_filters_pushdown
    (stream_schema_provider.rs:759-777) + expr_in_boundary (:960-976):
    a filter is Exact only when it is a minute-aligned comparison
    (>, >=, <, <=) on p_timestamp — those are fully answered by the
    minute-long partition prefixes; everything else is Inexact (row-group
    pruning + in-scan evaluation). A BETWEEN on p_timestamp is the injected
    two-bound conjunction (query/mod.rs:829-888): Exact iff both bounds are
    minute-aligned. NOTE: the gpuq engine evaluates EVERY pushed predicate
    on the GPU, so results are exact either way — this surface exists for a
    Rust ExecutionPlan shim that must answer DataFusion's planner
    (INTEGRATION.md)."""
    out = []
    for p in preds:
        op = p["op"]
        if p.get("col") == "p_timestamp":
            lit = p.get("lit")
            if (op in ("gt", "ge", "lt", "le")
                    and isinstance(lit, int) and not isinstance(lit, bool)
                    and lit % 60_000 == 0):
                out.append("exact")
                continue
            if (op == "between" and p["lo"] % 60_000 == 0
                    and p["hi"] % 60_000 == 0):
                out.append("exact")
                continue
        out.append("inexact")
    return out


class EmptyScanResult:
    """All files pruned at planning time: the scan is an empty relation —
    the aggregate result over zero rows (DataFusion's EmptyExec analog)."""

    def __init__(self, query):
        self.query = query

    def rows(self):
        return merge_partials([], self.query)


class ManifestCountResult:
    """Count fast path: bare `SELECT count(*)` with no value filters is
    answered from manifest num_rows sums and never reaches the scan engine
    (query.rs:189-256, query/mod.rs:537-590)."""

    def __init__(self, count):
        self.count = count

    def rows(self):
        return [[self.count]]


class StagingScanner:
    """The staging-window leg: `.arrows` Arrow-IPC files the
    ingest path has staged but not yet converted (read newest-first, the
    reverse reader of parseable/staging/reader.rs:316+). CPU evaluation of
    this in-memory slice is acceptable per §8f-2 and mirrors the reference's
    MemTable leg (stream_schema_provider.rs:292-350); it is NOT a fallback
    for the parquet hot path — staging .parquet files join the GPU plan."""

    def __init__(self, staging_dir: str):
        self.staging_dir = staging_dir

    def parquet_paths(self):
        import glob as _glob

        return sorted(_glob.glob(os.path.join(self.staging_dir, "*.parquet")))

    def _tables(self):
        import glob as _glob
        import pyarrow as pa

        for p in sorted(_glob.glob(os.path.join(self.staging_dir, "*.arrows")),
                        reverse=True):
            with pa.ipc.open_stream(p) as r:
                yield r.read_all()

    @staticmethod
    def _mask(tbl, query):
        import pyarrow as pa
        import pyarrow.compute as pc

        mask = pa.array([True] * tbl.num_rows)
        tr = query.get("time_range")
        preds = list(query.get("preds", []))
        if tr is not None:
            preds.append({"col": "p_timestamp", "op": "between",
                          "lo": tr[0], "hi": tr[1], "hi_exclusive": True})
        for p in preds:
            col = tbl.column(p["col"])
            if pa.types.is_dictionary(col.type):
                col = col.cast(col.type.value_type)
            if pa.types.is_timestamp(col.type):
                col = col.cast(pa.int64())
            op = p["op"]
            if op == "between":
                m = pc.and_(pc.greater_equal(col, p["lo"]),
                            pc.less(col, p["hi"]) if p.get("hi_exclusive")
                            else pc.less_equal(col, p["hi"]))
            elif op == "contains":
                m = pc.match_substring(col, p["lit"])
            else:
                f = {"eq": pc.equal, "ne": pc.not_equal, "lt": pc.less,
                     "le": pc.less_equal, "gt": pc.greater,
                     "ge": pc.greater_equal}[op]
                m = f(col, p["lit"])
            mask = pc.and_(mask, pc.fill_null(m, False))
        return mask

    def partial_batch(self, query):
        """-> pyarrow.RecordBatch in the partial-aggregate schema
        [key..., __presence, agg_i, agg_i_count, ...] or None."""
        import pyarrow as pa
        import pyarrow.compute as pc

        group_by = query.get("group_by", [])
        aggs = query["select"]
        acc = {}
        for tbl in self._tables():
            sel = tbl.filter(self._mask(tbl, query))
            if sel.num_rows == 0:
                continue
            keys = []
            for g in group_by:
                if isinstance(g, dict):  # DATE_BIN pseudo-key
                    ts = sel.column(g["bin"]).cast(pa.int64())
                    stride, origin = g["stride_ms"], g.get("origin", 0)
                    b = pc.add(pc.multiply(pc.floor(pc.divide(
                        pc.subtract(ts, origin), stride)).cast(pa.int64()),
                        stride), origin)
                    keys.append(b.to_pylist())
                else:
                    c = sel.column(g)
                    if pa.types.is_dictionary(c.type):
                        c = c.cast(c.type.value_type)
                    keys.append(c.to_pylist())
            vals = []
            for a in aggs:
                if a["agg"] == "count_star":
                    vals.append(None)
                else:
                    c = sel.column(a["col"])
                    if pa.types.is_dictionary(c.type):
                        c = c.cast(c.type.value_type)
                    if pa.types.is_timestamp(c.type):
                        c = c.cast(pa.int64())
                    vals.append(c.to_pylist())
            for r in range(sel.num_rows):
                key = tuple(k[r] for k in keys)
                st = acc.get(key)
                if st is None:
                    st = [0, [None] * len(aggs), [0] * len(aggs)]
                    acc[key] = st
                st[0] += 1
                for i, a in enumerate(aggs):
                    if a["agg"] == "count_star":
                        st[2][i] += 1
                        continue
                    v = vals[i][r]
                    if v is None:
                        continue
                    st[2][i] += 1
                    if a["agg"] == "count":
                        continue
                    cur = st[1][i]
                    if cur is None:
                        st[1][i] = v
                    elif a["agg"] in ("sum", "avg"):
                        st[1][i] = cur + v
                    elif a["agg"] == "min":
                        st[1][i] = min(cur, v)
                    elif a["agg"] == "max":
                        st[1][i] = max(cur, v)
        if not acc:
            return None
        nk = len(group_by)
        ks = sorted(acc.keys(), key=lambda k: tuple(
            ((1, "") if v is None else (0, v)) for v in k))
        cols, names = [], []
        for i in range(nk):
            names.append(f"k{i}")
            cols.append(pa.array([k[i] for k in ks]))
        names.append("__presence")
        cols.append(pa.array([acc[k][0] for k in ks], type=pa.int64()))
        for i, a in enumerate(aggs):
            names.append(f"a{i}")
            if a["agg"] in ("count_star", "count"):
                cols.append(pa.array([acc[k][2][i] for k in ks], type=pa.int64()))
            else:
                cols.append(pa.array([acc[k][1][i] for k in ks]))
            names.append(f"a{i}_count")
            cols.append(pa.array([acc[k][2][i] for k in ks], type=pa.int64()))
        return pa.record_batch(cols, names=names)

    def matching_rows(self, query):
        """Projection leg: matching staging rows [select_cols...]."""
        import pyarrow as pa

        out = []
        cols_wanted = query["select_cols"]
        for tbl in self._tables():
            sel = tbl.filter(self._mask(tbl, query))
            if sel.num_rows == 0:
                continue
            proj = []
            for cname in cols_wanted:
                c = sel.column(cname)
                if pa.types.is_dictionary(c.type):
                    c = c.cast(c.type.value_type)
                if pa.types.is_timestamp(c.type):
                    c = c.cast(pa.int64())
                proj.append(c.to_pylist())
            out.extend([list(r) for r in zip(*proj)])
        return out


class StagedPlan:
    """UnionExec analog over the GPU plan (manifest + staging parquet) and
    the CPU staging-arrows leg (stream_schema_provider.rs:384-401,637-647)."""

    def __init__(self, gpu_plan, scanner: StagingScanner, query: dict):
        self.gpu_plan = gpu_plan   # None when every parquet file was pruned
        self.scanner = scanner
        self.query = query

    def partition_count(self):
        return self.gpu_plan.partition_count() if self.gpu_plan else 0

    def load(self, partition=None):
        if self.gpu_plan:
            self.gpu_plan.load(partition)

    def execute(self, partition):
        return self.gpu_plan.execute(partition)

    def execute_all(self):
        batches = []
        if self.gpu_plan:
            batches = [self.gpu_plan.execute(p)
                       for p in range(self.gpu_plan.partition_count())]
            batches = [b for b in batches if b is not None]
        if self.query.get("select_cols"):
            return merge_topk(batches, self.query,
                              extra_rows=self.scanner.matching_rows(self.query))
        sb = self.scanner.partial_batch(self.query)
        if sb is not None:
            batches.append(sb)
        return merge_partials(batches, self.query)

    def metrics(self):
        return self.gpu_plan.metrics() if self.gpu_plan else {}

    def close(self):
        if self.gpu_plan:
            self.gpu_plan.close()


class StandardTableProvider:
    def __init__(self, stream_dir: str, session: GpuSession | None = None,
                 staging_dir: str | None = None, now_ms: int | None = None,
                 staging_window_ms: int = 5 * 60_000):
        self.stream_dir = stream_dir
        self.session = session
        self.staging_dir = staging_dir
        self.now_ms = now_ms
        self.staging_window_ms = staging_window_ms
        snap_path = os.path.join(stream_dir, "stream.json")
        with open(snap_path) as fh:
            self.stream_json = json.load(fh)
        self.snapshot = self.stream_json["snapshot"]

    # is_within_staging_window (stream_schema_provider.rs:936-958): the query
    # range touches staging when its upper bound reaches now - window (or
    # there is no range / no clock). The reference truncates the boundary to
    # the minute (with_second(0).with_nanosecond(0)) and compares with >=.
    def _staging_touches(self, time_range):
        if self.staging_dir is None:
            return False
        if time_range is None or self.now_ms is None:
            return True
        boundary = (self.now_ms - self.staging_window_ms) // 60_000 * 60_000
        return time_range[1] >= boundary

    # Legacy listing branch (SURVEY §2 (★); listing_table_builder.rs:46-118
    # + is_overlapping_query / return_listing_time_filters,
    # stream_schema_provider.rs:672-689,845-930): data older than the first
    # manifest is discovered by minute-granularity prefix listing
    # (date=/hour=/minute=) and joins the same scan plan — per-row time
    # filters stay exact on the engine side.
    def _first_manifest_lower(self):
        items = self.snapshot["manifest_list"]
        if not items:
            return None
        return min(_parse_iso_ms(it["time_lower_bound"]) for it in items)

    def _legacy_listing_files(self, lo_ms, hi_ms):
        from datetime import datetime, timezone

        files = []
        minute = lo_ms // 60_000 * 60_000
        while minute < hi_ms:
            dt = datetime.fromtimestamp(minute / 1000, tz=timezone.utc)
            prefix = os.path.join(self.stream_dir, f"date={dt:%Y-%m-%d}",
                                  f"hour={dt:%H}", f"minute={dt:%M}")
            if os.path.isdir(prefix):
                files.extend(
                    os.path.join(prefix, f) for f in os.listdir(prefix)
                    if f.endswith(".parquet"))
            minute += 60_000
        files.sort(reverse=True)  # listing.sorted().rev()
        return files

    def _legacy_files(self, time_range):
        """Files for the pre-manifest slice of the query range, or []."""
        first_lower = self._first_manifest_lower()
        if first_lower is None:
            # no manifests at all: everything rides the listing — an explicit
            # range is required (listing_table_builder.rs:79-84)
            if time_range is None:
                raise GpuqError(
                    "time predicate required to query pre-manifest data "
                    "(listing_table_builder.rs:79-84)")
            return self._legacy_listing_files(time_range[0], time_range[1])
        if time_range is None or time_range[0] >= first_lower:
            return []  # is_overlapping_query == false
        return self._legacy_listing_files(time_range[0],
                                          min(time_range[1], first_lower))

    # Snapshot::manifests (catalog/snapshot.rs:42-71): retain manifests whose
    # [lower,upper] overlaps the time predicates.
    def _select_manifests(self, time_range):
        items = self.snapshot["manifest_list"]
        if time_range:
            lo, hi = time_range
            items = [
                it
                for it in items
                if _parse_iso_ms(it["time_upper_bound"]) >= lo
                and _parse_iso_ms(it["time_lower_bound"]) < hi
            ]
        return items

    def _manifest_files(self, time_range):
        files = []
        for it in self._select_manifests(time_range):
            mpath = it["manifest_path"]
            if not os.path.isabs(mpath):
                mpath = os.path.join(os.path.dirname(self.stream_dir), mpath)
            files.extend(_load_manifest(mpath)["files"])
        return files

    def scan(self, query: dict):
        """query: the IR of tests/golden_queries.py (select/group_by/preds/
        time_range). Returns GpuExecutionPlan, StagedPlan (staging window),
        ManifestCountResult or EmptyScanResult."""
        query = {k: v for k, v in query.items() if k != "ext"}

        staging_hit = self._staging_touches(query.get("time_range"))
        if staging_hit:
            return self._scan_with_staging(query)

        legacy_paths = self._legacy_files(query.get("time_range"))

        if legacy_paths:
            return self._scan_with_legacy(query, legacy_paths)

        if self.session is not None and not os.environ.get("GPUQ_PY_PLANNER"):
            # native catalog planner inside libgpuq (catalog.cpp)
            plan = GpuExecutionPlan(self.session, None, query,
                                    stream_dir=self.stream_dir)
            if plan.fast_count is not None:
                return ManifestCountResult(plan.fast_count)
            if plan.empty:
                return EmptyScanResult(query)
            return plan

        preds = list(query.get("preds", []))
        time_range = query.get("time_range")
        files = self._manifest_files(time_range)

        # prune: time bounds as predicates on p_timestamp + value predicates
        prune_preds = preds.copy()
        if time_range:
            prune_preds.append(
                {
                    "col": "p_timestamp",
                    "op": "between",
                    "lo": time_range[0],
                    "hi": time_range[1],
                    "hi_exclusive": True,
                }
            )
        kept = [f for f in files if not _file_pruned(f, prune_preds)]

        # count fast path (exact only when no value preds and the time range
        # either is absent or fully covers every kept file's ts bounds)
        sel = query.get("select", [])
        if (
            not preds
            and not query.get("group_by")
            and not query.get("select_cols")
            and len(sel) == 1
            and sel[0]["agg"] == "count_star"
        ):
            exact = True
            if time_range:
                for fe in kept:
                    st = _stats_entry(
                        {c["name"]: c for c in fe["columns"]}.get("p_timestamp")
                    )
                    if not st or "Int" not in st:
                        exact = False
                        break
                    mm = st["Int"]
                    if not (time_range[0] <= mm["min"] and mm["max"] < time_range[1]):
                        exact = False
                        break
            if exact:
                return ManifestCountResult(sum(f["num_rows"] for f in kept))

        if not kept:
            return EmptyScanResult(query)
        paths = []
        root = os.path.dirname(self.stream_dir)
        for fe in kept:
            p = fe["file_path"]
            paths.append(p if os.path.isabs(p) else os.path.join(root, p))
        return GpuExecutionPlan(self.session, paths, query)

    def _scan_with_staging(self, query):
        """Staging branch (stream_schema_provider.rs:637-647): manifest
        parquet + staging parquet execute on the GPU; staging .arrows run on
        the CPU leg; the manifest count fast path is SKIPPED because staging
        rows are not in any manifest yet."""
        scanner = StagingScanner(self.staging_dir)
        preds = list(query.get("preds", []))
        time_range = query.get("time_range")
        files = self._manifest_files(time_range)
        prune_preds = preds.copy()
        if time_range:
            prune_preds.append({"col": "p_timestamp", "op": "between",
                                "lo": time_range[0], "hi": time_range[1],
                                "hi_exclusive": True})
        kept = [f for f in files if not _file_pruned(f, prune_preds)]
        root = os.path.dirname(self.stream_dir)
        paths = []
        for fe in kept:
            p = fe["file_path"]
            paths.append(p if os.path.isabs(p) else os.path.join(root, p))
        paths.extend(scanner.parquet_paths())  # scanned unconditionally
        gpu_plan = GpuExecutionPlan(self.session, paths, query) if paths else None
        return StagedPlan(gpu_plan, scanner, query)

    def _scan_with_legacy(self, query, legacy_paths):
        """Pre-manifest files join the manifested scan: legacy files carry
        no manifest stats (only footer stats prune them), and the count fast
        path is skipped — their row counts are in no manifest
        (stream_schema_provider.rs:672-689)."""
        preds = list(query.get("preds", []))
        time_range = query.get("time_range")
        files = self._manifest_files(time_range)
        prune_preds = preds.copy()
        if time_range:
            prune_preds.append({"col": "p_timestamp", "op": "between",
                                "lo": time_range[0], "hi": time_range[1],
                                "hi_exclusive": True})
        kept = [f for f in files if not _file_pruned(f, prune_preds)]
        root = os.path.dirname(self.stream_dir)
        paths = list(legacy_paths)
        for fe in kept:
            p = fe["file_path"]
            paths.append(p if os.path.isabs(p) else os.path.join(root, p))
        if not paths:
            return EmptyScanResult(query)
        return GpuExecutionPlan(self.session, paths, query)


def _build_c_projection(query: dict, keep):
    cols = query.get("select_cols", [])
    cb = [c.encode() for c in cols]
    keep.extend(cb)
    arr = (C.c_char_p * max(len(cols), 1))(*cb)
    limit = int(query.get("limit", -1))  # < 0: unlimited (streamed batches)
    ob = query.get("order_by")
    if cols:
        if ob and not (ob.get("col") == "p_timestamp" and ob.get("desc", True)):
            raise GpuqError("only ORDER BY p_timestamp DESC is supported "
                            "(the reference's output-ordering contract, "
                            "stream_schema_provider.rs:181-204)")
        if "p_timestamp" not in cols:
            raise GpuqError("projection must include p_timestamp (merge key)")
    return arr, len(cols), limit


def _build_c_query(query: dict, keep):
    """query IR -> (cpreds, n_preds, cgroup, n_group, caggs, n_aggs).
    `keep` collects byte strings that must outlive the C call."""
    preds = list(query.get("preds", []))
    tr = query.get("time_range")
    if tr:
        preds.append({"col": "p_timestamp", "op": "between",
                      "lo": tr[0], "hi": tr[1], "hi_exclusive": True})
    cpreds = (GpuqPred * max(len(preds), 1))()
    for i, p in enumerate(preds):
        b = p["col"].encode()
        keep.append(b)
        cpreds[i].column = b
        cpreds[i].op = OPS[p["op"]]
        cpreds[i].hi_exclusive = 1 if p.get("hi_exclusive") else 0
        if p["op"] == "between":
            cpreds[i].lit_kind = 0
            cpreds[i].i64[0] = p["lo"]
            cpreds[i].i64[1] = p["hi"]
        else:
            lit = p["lit"]
            if isinstance(lit, str):
                cpreds[i].lit_kind = 2
                lb = lit.encode()
                keep.append(lb)
                cpreds[i].str = lb
            elif isinstance(lit, float):
                cpreds[i].lit_kind = 1
                cpreds[i].f64[0] = lit
            else:
                cpreds[i].lit_kind = 0
                cpreds[i].i64[0] = lit
    group_by = query.get("group_by", [])
    gbb = []
    for g in group_by:
        if isinstance(g, dict):  # DATE_BIN pseudo-key (query/mod.rs:665-735)
            g = f"__bin:{g['bin']}:{g['stride_ms']}:{g.get('origin', 0)}"
        gbb.append(g.encode())
    keep.extend(gbb)
    cgroup = (C.c_char_p * max(len(group_by), 1))(*gbb)
    aggs = query.get("select", [])
    caggs = (GpuqAgg * max(len(aggs), 1))()
    for i, a in enumerate(aggs):
        # avg = sum/count at the Final merge (DataFusion's Avg partial state
        # is (sum, count) too); the engine only ever computes the exact sum
        caggs[i].op = AGGS["sum" if a["agg"] == "avg" else a["agg"]]
        if a.get("col"):
            ab = a["col"].encode()
            keep.append(ab)
            caggs[i].column = ab
    return cpreds, len(preds), cgroup, len(group_by), caggs, len(aggs)


class GpuExecutionPlan:
    def __init__(self, session: GpuSession, paths: list[str] | None, query: dict,
                 stream_dir: str | None = None):
        if session is None:
            raise GpuqError("a GpuSession (GPU) is required for scan execution")
        self.session = session
        self.query = query
        lib = session._lib
        self._lib = lib
        self._keep = []
        self.fast_count = None   # set by the native-planner path
        self.empty = False

        cpreds, np_, cgroup, ng, caggs, na = _build_c_query(query, self._keep)
        cproj, nproj, limit = _build_c_projection(query, self._keep)
        if stream_dir is not None:
            # native catalog planner (§8f row 1): manifest selection, pruning
            # and the count fast path run inside libgpuq (catalog.cpp)
            fc = C.c_int64(-1)
            self._plan = lib.gpuq_plan_build_from_stream(
                session._ctx, stream_dir.encode(),
                cproj, nproj,
                cpreds, np_, cgroup, ng, caggs, na,
                C.c_int64(limit), C.byref(fc))
            if not self._plan:
                if fc.value >= 0:
                    self.fast_count = fc.value
                    return
                if fc.value == -2:
                    self.empty = True
                    return
                raise GpuqError(f"plan_build_from_stream failed: {session._err()}")
            return

        n = len(paths)
        self._path_bytes = [p.encode() for p in paths]
        files = (GpuqFile * max(n, 1))()
        for i, pb in enumerate(self._path_bytes):
            files[i].path = pb
            files[i].row_groups = None
            files[i].n_row_groups = -1
        self._plan = lib.gpuq_plan_build(
            session._ctx,
            files, n,
            cproj, nproj,
            cpreds, np_,
            cgroup, ng,
            caggs, na,
            C.c_int64(limit),
        )
        if not self._plan:
            raise GpuqError(f"plan_build failed: {session._err()}")

    def partition_count(self) -> int:
        return self._lib.gpuq_plan_partition_count(self._plan)

    def load(self, partition=None):
        parts = range(self.partition_count()) if partition is None else [partition]
        for p in parts:
            if self._lib.gpuq_plan_load(self._plan, p) != 0:
                raise GpuqError(f"plan_load failed: {self.session._err()}")

    def execute_reader(self, partition: int):
        """-> pyarrow.RecordBatchReader over this partition's result: one
        partial-aggregate batch for aggregations, or a true multi-batch
        stream of 20k-row batches for (unlimited) projection scans —
        ExecutionPlan::execute's pull-based stream (query/mod.rs:341-368)."""
        import pyarrow as pa

        # allocate an ArrowArrayStream struct (5 ptr fields + private)
        buf = C.create_string_buffer(6 * C.sizeof(C.c_void_p))
        rc = self._lib.gpuq_plan_execute(self._plan, partition, C.cast(buf, C.c_void_p))
        if rc != 0:
            raise GpuqError(f"plan_execute failed: {self.session._err()}")
        return pa.RecordBatchReader._import_from_c(C.addressof(buf))

    def execute(self, partition: int):
        """Drained form of execute_reader: one combined RecordBatch."""
        import pyarrow as pa

        batches = list(self.execute_reader(partition))
        if len(batches) <= 1:
            return batches[0] if batches else None
        t = pa.Table.from_batches(batches).combine_chunks()
        return t.to_batches()[0] if t.num_rows else None

    def execute_all(self):
        """Run every partition and apply the Final merge: aggregation
        (AggregateExec(Final)) or top-k row merge for projection scans."""
        batches = [self.execute(p) for p in range(self.partition_count())]
        batches = [b for b in batches if b is not None]
        if self.query.get("select_cols"):
            return merge_topk(batches, self.query)
        return merge_partials(batches, self.query)

    def metrics(self) -> dict:
        m = GpuqMetrics()
        if self._lib.gpuq_plan_metrics(self._plan, C.byref(m)) != 0:
            raise GpuqError("plan_metrics failed")
        return {f: getattr(m, f) for f, _ in m._fields_}

    def close(self):
        if getattr(self, "_plan", None):
            self._lib.gpuq_plan_destroy(self._plan)
            self._plan = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def _col_list(col):
    """Fast column -> python list: to_pylist() costs ~1 us/element (it
    dominated the per-step Final merge at 1000 groups); the numpy route is
    20-50x faster. Null-bearing columns keep to_pylist (to_numpy would
    coerce nulled ints to floats)."""
    if col.null_count == 0:
        return col.to_numpy(zero_copy_only=False).tolist()
    return col.to_pylist()


def merge_partials(batches, query):
    """Final merge of partial aggregate tables (cross-partition and
    cross-rank). Output rows match the oracle's normalized form:
    [key..., agg...] sorted by key tuple, NULL keys last."""
    group_by = query.get("group_by", [])
    aggs = query["select"]
    nk = len(group_by)
    if len(batches) == 1 and batches[0] is not None and batches[0].num_rows:
        # a single partial IS the final result: the exported value columns
        # already hold final values (counts in the value slot for count
        # aggregates, NULL validity where count==0) — pure column zips,
        # no per-row Python. The partial arrives key-sorted from the C
        # export (live-group order == key order only for single-key
        # plans), so sort via numpy lexsort on key codes instead of a
        # python tuple sort.
        import pyarrow as pa
        import pyarrow.compute as pc

        b = batches[0]
        picked_idx = list(range(nk)) + [nk + 1 + 2 * i
                                        for i in range(len(aggs))]
        cols = [_col_list(b.column(i)) for i in picked_idx]
        for i, a in enumerate(aggs):
            if a["agg"] != "avg":
                continue
            cnts = b.column(nk + 2 + 2 * i).to_pylist()
            cols[nk + i] = [
                (None if not c else s / c)
                for s, c in zip(cols[nk + i], cnts)
            ]
        if nk:
            kt = pa.table({f"k{i}": b.column(i) for i in range(nk)})
            # default null_placement is at_end — matches the oracle's
            # NULLs-last normalized ordering
            order = pc.sort_indices(
                kt, sort_keys=[(f"k{i}", "ascending")
                               for i in range(nk)]).to_pylist()
            return [[c[i] for c in cols] for i in order]
        return [list(t) for t in zip(*cols)]
    acc = {}
    for b in batches:
        if b is None or b.num_rows == 0:
            continue
        cols = [_col_list(b.column(i)) for i in range(b.num_columns)]
        n = b.num_rows
        for r in range(n):
            key = tuple(cols[k][r] for k in range(nk))
            presence = cols[nk][r]
            st = acc.get(key)
            if st is None:
                st = {"presence": 0, "vals": [None] * len(aggs), "cnts": [0] * len(aggs)}
                acc[key] = st
            st["presence"] += presence
            for i, a in enumerate(aggs):
                v = cols[nk + 1 + 2 * i][r]
                c = cols[nk + 2 + 2 * i][r]
                if a["agg"] in ("count_star", "count"):
                    st["cnts"][i] += v
                    continue
                if c == 0 or v is None:
                    continue
                st["cnts"][i] += c
                if st["vals"][i] is None:
                    st["vals"][i] = v
                elif a["agg"] in ("sum", "avg"):
                    st["vals"][i] += v
                elif a["agg"] == "min":
                    st["vals"][i] = min(st["vals"][i], v)
                elif a["agg"] == "max":
                    st["vals"][i] = max(st["vals"][i], v)

    rows = []
    for key, st in acc.items():
        if st["presence"] == 0 and group_by:
            continue
        row = list(key)
        for i, a in enumerate(aggs):
            if a["agg"] in ("count_star", "count"):
                row.append(st["cnts"][i])
            elif a["agg"] == "avg":
                row.append(st["vals"][i] / st["cnts"][i]
                           if st["cnts"][i] > 0 else None)
            else:
                row.append(st["vals"][i] if st["cnts"][i] > 0 else None)
        rows.append(row)
    rows.sort(key=lambda r: tuple(((1, "") if v is None else (0, v)) for v in r[: len(group_by)]))
    if not group_by and not rows:
        rows = [[0 if a["agg"] in ("count_star", "count") else None for a in aggs]]
    return rows


def merge_topk(batches, query, extra_rows=None):
    """Final merge of per-partition projection batches: re-sort by
    p_timestamp DESC across partitions, truncate to LIMIT when one is set
    (unlimited scans return every matching row). extra_rows:
    already-projected rows from the staging CPU leg."""
    cols = query["select_cols"]
    ts_i = cols.index("p_timestamp")
    rows = list(extra_rows) if extra_rows else []
    for b in batches:
        if b is None or b.num_rows == 0:
            continue
        data = [_col_list(b.column(i)) for i in range(b.num_columns)]
        rows.extend([list(r) for r in zip(*data)])
    rows.sort(key=lambda r: -r[ts_i])
    lim = query.get("limit")
    return rows[: int(lim)] if lim else rows


class Query:
    """The single execution funnel, mirroring Query::execute
    (query/mod.rs:152-165,291-372)."""

    def __init__(self, provider: StandardTableProvider):
        self.provider = provider

    def execute(self, query: dict):
        plan = self.provider.scan(query)
        if isinstance(plan, (ManifestCountResult, EmptyScanResult)):
            return plan.rows(), None
        try:
            rows = plan.execute_all()
            metrics = plan.metrics()
            return rows, metrics
        finally:
            plan.close()
