"""CPU oracle #1: pyarrow-decode + numpy restatement of the reference's
query semantics for the hot path (SELECT [keys,] aggs FROM stream WHERE
<conjunction> GROUP BY keys).

ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py header).

Semantics restated from the reference and its engine (DataFusion 53.1 /
arrow 58.1, pinned in /root/reference/Cargo.lock — those crates are not
vendored under /root/reference, so this restates their *published* SQL
semantics, anchored on the reference's call sites and tests):
  - time-range filter injected on every scan as
    `p_timestamp >= start AND p_timestamp < end` (ms)
                                  -- src/query/mod.rs:829-888 (`transform`)
  - count(*) counts rows incl. nulls; count(col) counts non-null
  - a predicate never matches NULL (SQL three-valued logic)
  - GROUP BY keeps a NULL group
  - min/max over utf8 = lexicographic byte order
  - sum(int64) accumulates in i64; BETWEEN lo AND hi inclusive both ends
  - LIKE '%x%' = byte substring containment
  - integer results bit-exact; float sums within 1 ULP (BASELINE.json gate)

Group output order is engine-defined (hash agg); results are normalized to
key-sorted rows — the reference pins no output order without ORDER BY.

Pinned against (tests/test_oracle.py): (a) pyarrow Acero running the same
plan (execute_acero below — an independent vectorized engine of the same
family as the reference's), and (b) the independent scalar C restatement of
the full path including parquet decode (oracle/cpu_ref.c).
"""

from __future__ import annotations

import numpy as np
import pyarrow as pa
import pyarrow.compute as pc
import pyarrow.parquet as pq

I64_MIN = -(2**63)
I64_MAX = 2**63 - 1


def _needed_columns(query: dict) -> set[str]:
    need = set()
    for g in query.get("group_by", []):
        need.add(g["bin"] if isinstance(g, dict) else g)
    for p in query.get("preds", []):
        need.add(p["col"])
    for s in query["select"]:
        if s.get("col"):
            need.add(s["col"])
    if query.get("time_range"):
        need.add("p_timestamp")
    return need


def _norm_col(col) -> pa.Array:
    col = col.combine_chunks() if isinstance(col, pa.ChunkedArray) else col
    if pa.types.is_timestamp(col.type):
        col = col.cast(pa.int64())
    return col


def _pred_mask(pred: dict, col: pa.Array) -> np.ndarray:
    """Predicate mask with SQL semantics (NULL never matches).
    Mirrors arrow-rs compute kernels the reference's FilterExec uses."""
    op = pred["op"]
    if op == "between":
        e = pc.and_(pc.greater_equal(col, pred["lo"]), pc.less_equal(col, pred["hi"]))
    elif op == "contains":
        if pa.types.is_dictionary(col.type):
            col = col.cast(pa.string())
        e = pc.match_substring(col, pred["lit"])
    else:
        if pa.types.is_dictionary(col.type):
            col = col.cast(col.type.value_type)
        f = {
            "eq": pc.equal,
            "ne": pc.not_equal,
            "lt": pc.less,
            "le": pc.less_equal,
            "gt": pc.greater,
            "ge": pc.greater_equal,
        }[op]
        e = f(col, pred["lit"])
    e = e.fill_null(False)
    return np.asarray(e)


def _factorize(col: pa.Array):
    """-> (codes int64 ndarray with -1 for NULL, values: list)."""
    if pa.types.is_dictionary(col.type):
        enc = col
    else:
        enc = col.dictionary_encode()
    codes = enc.indices.fill_null(-1).to_numpy(zero_copy_only=False).astype(np.int64)
    return codes, enc.dictionary.to_pylist()


def _sources(files, extra_tables, need):
    """Yield pyarrow Tables: parquet files (column-pruned reads) + in-memory
    tables (the staging .arrows slice of SURVEY §8f-2)."""
    for path in files:
        yield pq.read_table(path, columns=need or None)
    for t in extra_tables or []:
        yield t.select(need) if need else t


def execute_projection(files: list[str], query: dict,
                       extra_tables=None) -> dict:
    """Projection scan: SELECT cols WHERE ... ORDER BY p_timestamp DESC
    LIMIT k (the console default; ordering contract
    stream_schema_provider.rs:181-204). Ties at the LIMIT boundary are
    engine-defined — compare tie-tolerantly (see tests)."""
    cols_wanted = query["select_cols"]
    need = sorted(set(cols_wanted)
                  | {p["col"] for p in query.get("preds", [])}
                  | {"p_timestamp"})
    out = []
    for tbl in _sources(files, extra_tables, need):
        n = tbl.num_rows
        cols = {name: _norm_col(tbl.column(name)) for name in need}
        mask = np.ones(n, dtype=bool)
        tr = query.get("time_range")
        if tr is not None:
            ts = np.asarray(cols["p_timestamp"])
            mask &= (ts >= tr[0]) & (ts < tr[1])
        for p in query.get("preds", []):
            mask &= _pred_mask(p, cols[p["col"]])
        sel = np.nonzero(mask)[0]
        proj = []
        for cname in cols_wanted:
            c = cols[cname]
            if pa.types.is_dictionary(c.type) or pa.types.is_string(c.type):
                vals = (c.cast(pa.string()) if pa.types.is_dictionary(c.type) else c).to_pylist()
                proj.append([vals[i] for i in sel])
            else:
                arr = c.to_pylist()
                proj.append([arr[i] for i in sel])
        out.extend([list(r) for r in zip(*proj)])
    ts_i = cols_wanted.index("p_timestamp")
    out.sort(key=lambda r: -r[ts_i])
    lim = query.get("limit")
    return {"columns": cols_wanted,
            "rows": out[: int(lim)] if lim else out,
            "all_matching": out}


def execute(files: list[str], query: dict, extra_tables=None) -> dict:
    """Run the query. Returns {"columns": [...], "rows": [[key...,agg...]...]}
    rows sorted by key tuple, NULLs last. extra_tables: in-memory pyarrow
    Tables included as additional sources (staging .arrows)."""
    if query.get("select_cols"):
        return execute_projection(files, query, extra_tables)
    need = sorted(_needed_columns(query))
    group_by = query.get("group_by", [])
    aggs = query["select"]
    acc: dict[tuple, list] = {}

    for tbl in _sources(files, extra_tables, need):
        n = tbl.num_rows
        cols = {name: _norm_col(tbl.column(name)) for name in need}
        mask = np.ones(n, dtype=bool)
        tr = query.get("time_range")
        if tr is not None:
            ts = np.asarray(cols["p_timestamp"])
            mask &= (ts >= tr[0]) & (ts < tr[1])  # query/mod.rs:829-888
        for p in query.get("preds", []):
            mask &= _pred_mask(p, cols[p["col"]])
        sel = np.nonzero(mask)[0]
        if len(sel) == 0:
            continue

        # factorize group keys -> combined code per selected row
        if group_by:
            codes_list, values_list = [], []
            for g in group_by:
                if isinstance(g, dict):
                    # DATE_BIN: key = origin-aligned bin start (ms)
                    ts = np.asarray(cols[g["bin"]])
                    stride, origin = g["stride_ms"], g.get("origin", 0)
                    bins = (ts - origin) // stride * stride + origin
                    uniq_b = np.unique(bins)
                    lookup = {int(b): i for i, b in enumerate(uniq_b.tolist())}
                    c = np.array([lookup[int(b)] for b in bins], dtype=np.int64)
                    v = [int(b) for b in uniq_b.tolist()]
                else:
                    c, v = _factorize(cols[g])
                codes_list.append(c[sel] + 1)  # 0 = NULL group
                values_list.append([None] + v)
            combined = codes_list[0]
            sizes = [len(v) for v in values_list]
            for c, s in zip(codes_list[1:], sizes[1:]):
                combined = combined * s + c
            uniq, inv = np.unique(combined, return_inverse=True)
            # decode uniq back to key tuples
            key_tuples = []
            for u in uniq.tolist():
                parts = []
                for s, v in zip(reversed(sizes), reversed(values_list)):
                    parts.append(v[u % s])
                    u //= s
                key_tuples.append(tuple(reversed(parts)))
        else:
            uniq = np.zeros(1)
            inv = np.zeros(len(sel), dtype=np.int64)
            key_tuples = [()]
        G = len(key_tuples)

        # per-file aggregates, vectorized
        file_res = []
        for a in aggs:
            op = a["agg"]
            if op == "count_star":
                file_res.append(("count", np.bincount(inv, minlength=G).astype(np.int64)))
                continue
            col = cols[a["col"]]
            valid = np.ones(n, dtype=bool)
            if col.null_count:
                valid = ~np.asarray(col.is_null())
            vsel = valid[sel]
            if op == "count":
                file_res.append(("count", np.bincount(inv[vsel], minlength=G).astype(np.int64)))
                continue
            if pa.types.is_string(col.type) or (
                pa.types.is_dictionary(col.type)
                and pa.types.is_string(col.type.value_type)
            ):
                # utf8 min/max: lexicographic byte order (DataFusion semantics)
                if op not in ("min", "max"):
                    raise NotImplementedError(f"{op} over utf8")
                scol = col.cast(pa.string()) if pa.types.is_dictionary(col.type) else col
                vals = scol.to_pylist()
                out = [None] * G
                cnt = np.zeros(G, dtype=np.int64)
                for j, i in enumerate(sel):
                    if not vsel[j]:
                        continue
                    v = vals[i]
                    g = inv[j]
                    cnt[g] += 1
                    if out[g] is None or (v < out[g] if op == "min" else v > out[g]):
                        out[g] = v
                file_res.append((op, out, cnt))
                continue
            vals = np.asarray(
                col.cast(col.type.value_type) if pa.types.is_dictionary(col.type) else col
            )
            fv = vals[sel][vsel]
            gi = inv[vsel]
            if op in ("sum", "avg"):
                is_f = np.issubdtype(fv.dtype, np.floating)
                if is_f:
                    # f64 sums: keep the per-group VALUES so the final sum is
                    # math.fsum — correctly rounded, matching the GPU's exact
                    # 256-bit superaccumulator within 1 ULP (BASELINE gate)
                    order = np.argsort(gi, kind="stable")
                    sgi, sfv = gi[order], fv[order]
                    bounds = np.searchsorted(sgi, np.arange(G + 1))
                    parts = [sfv[bounds[g]:bounds[g + 1]] for g in range(G)]
                    cnt = np.bincount(gi, minlength=G)
                    file_res.append(("fsum", parts, cnt))
                    continue
                out = np.zeros(G, dtype=np.int64)
                np.add.at(out, gi, fv)
                cnt = np.bincount(gi, minlength=G)
                file_res.append((op, out, cnt))  # "sum" or "avg" (i64)
            elif op == "min":
                out = np.full(G, np.inf if np.issubdtype(fv.dtype, np.floating) else I64_MAX,
                              dtype=fv.dtype if np.issubdtype(fv.dtype, np.floating) else np.int64)
                np.minimum.at(out, gi, fv)
                cnt = np.bincount(gi, minlength=G)
                file_res.append(("min", out, cnt))
            elif op == "max":
                out = np.full(G, -np.inf if np.issubdtype(fv.dtype, np.floating) else I64_MIN,
                              dtype=fv.dtype if np.issubdtype(fv.dtype, np.floating) else np.int64)
                np.maximum.at(out, gi, fv)
                cnt = np.bincount(gi, minlength=G)
                file_res.append(("max", out, cnt))
            else:
                raise ValueError(op)

        # merge this file's groups into the global accumulator (per-group cost)
        for gidx, k in enumerate(key_tuples):
            st = acc.get(k)
            if st is None:
                st = [None] * len(aggs)
                acc[k] = st
            for ai, r in enumerate(file_res):
                kind = r[0]
                if kind == "count":
                    v = int(r[1][gidx])
                    st[ai] = v if st[ai] is None else st[ai] + v
                elif kind == "fsum":
                    part = r[1][gidx]
                    if len(part):
                        if st[ai] is None:
                            st[ai] = []
                        st[ai].append(part)
                elif kind == "avg":  # i64 avg: carry (exact int sum, count)
                    c2 = int(r[2][gidx])
                    if c2:
                        v = int(r[1][gidx])
                        st[ai] = ((v, c2) if st[ai] is None
                                  else (st[ai][0] + v, st[ai][1] + c2))
                else:
                    v = r[1][gidx]
                    present = int(r[2][gidx]) > 0
                    if not present:
                        continue
                    if hasattr(v, "item"):
                        v = v.item()
                    if st[ai] is None:
                        st[ai] = v
                    elif kind == "sum":
                        st[ai] += v
                    elif kind == "min":
                        st[ai] = min(st[ai], v)
                    elif kind == "max":
                        st[ai] = max(st[ai], v)

    out_cols = list(group_by) + [_agg_name(a) for a in aggs]
    rows = []
    for k, st in acc.items():
        row = list(k)
        for a, s in zip(aggs, st):
            if s is None and a["agg"] in ("count", "count_star"):
                s = 0
            elif isinstance(s, list):  # f64 sum/avg: exact, rounded once
                import math

                allv = np.concatenate(s)
                s = math.fsum(allv)
                if a["agg"] == "avg":
                    s = s / len(allv)
            elif isinstance(s, tuple):  # i64 avg: exact sum / count
                s = s[0] / s[1]
            row.append(s)
        rows.append(row)
    rows.sort(key=lambda r: tuple(_sort_key(v) for v in r[: len(group_by)]))
    if not group_by and not rows:
        rows = [[0 if a["agg"] in ("count", "count_star") else None for a in aggs]]
    return {"columns": out_cols, "rows": rows}


def _sort_key(v):
    return (1, "") if v is None else (0, v)


def _agg_name(a):
    return a.get("name") or (
        "count(*)" if a["agg"] == "count_star" else f"{a['agg']}({a.get('col')})"
    )


# ---------------------------------------------------------------------------
# Oracle #1b — the same plan through pyarrow Acero (pyarrow.dataset +
# TableGroupBy). Used by tests to pin the numpy restatement against an
# independent engine, and by bench.py as the reported multicore CPU-baseline
# stand-in for the reference's CPU DataFusion path.
# ---------------------------------------------------------------------------

def _acero_filter(query: dict):
    import pyarrow.dataset as ds  # noqa: F401

    filt = None

    def AND(a, b):
        return b if a is None else (a & b)

    tr = query.get("time_range")
    if tr is not None:
        f = pc.field("p_timestamp")
        filt = AND(
            filt,
            (f >= pa.scalar(tr[0], pa.timestamp("ms")))
            & (f < pa.scalar(tr[1], pa.timestamp("ms"))),
        )
    for p in query.get("preds", []):
        f = pc.field(p["col"])
        op = p["op"]
        if op == "between":
            lo, hi = p["lo"], p["hi"]
            if p["col"] == "p_timestamp":
                lo = pa.scalar(lo, pa.timestamp("ms"))
                hi = pa.scalar(hi, pa.timestamp("ms"))
            e = (f >= lo) & (f <= hi)
        elif op == "contains":
            e = pc.match_substring(f, p["lit"])
        else:
            e = {
                "eq": f == p["lit"],
                "ne": f != p["lit"],
                "lt": f < p["lit"],
                "le": f <= p["lit"],
                "gt": f > p["lit"],
                "ge": f >= p["lit"],
            }[op]
        filt = AND(filt, e)
    return filt


def execute_acero(files: list[str], query: dict) -> dict:
    import pyarrow.dataset as ds

    dataset = ds.dataset(files, format="parquet")
    filt = _acero_filter(query)
    group_by = query.get("group_by", [])
    need = sorted(_needed_columns(query))
    tbl = dataset.to_table(filter=filt, columns=need or None)
    # decode dictionary key columns for grouping; timestamps compare as ms
    for g in group_by:
        i = tbl.schema.get_field_index(g)
        if pa.types.is_dictionary(tbl.schema.field(g).type):
            tbl = tbl.set_column(i, g, tbl.column(g).cast(pa.string()))
        elif pa.types.is_timestamp(tbl.schema.field(g).type):
            tbl = tbl.set_column(i, g, tbl.column(g).cast(pa.int64()))

    agglist = []
    names = []
    for a in query["select"]:
        op = a["agg"]
        if op == "count_star":
            agglist.append(([], "count_all"))
            names.append("count_all")
        else:
            aop = {"avg": "mean"}.get(op, op)
            agglist.append((a["col"], aop))
            names.append(f"{a['col']}_{aop}")

    if group_by:
        res = pa.TableGroupBy(tbl, group_by).aggregate(agglist)
    else:
        tbl2 = tbl.append_column("__g", pa.array(np.zeros(tbl.num_rows, dtype=np.int8)))
        res = pa.TableGroupBy(tbl2, ["__g"]).aggregate(agglist).drop_columns(["__g"])
    # normalize timestamp-typed aggregate results (min/max over
    # p_timestamp) to i64 ms like the numpy oracle
    for name in res.column_names:
        idx = res.schema.get_field_index(name)
        if pa.types.is_timestamp(res.schema.field(name).type):
            res = res.set_column(idx, name, res.column(name).cast(pa.int64()))
    aggcols = [c for c in res.column_names if c not in group_by]
    rows = []
    for i in range(res.num_rows):
        key = [res.column(c)[i].as_py() for c in group_by]
        vals = [res.column(c)[i].as_py() for c in aggcols]
        rows.append(key + vals)
    rows.sort(key=lambda r: tuple(_sort_key(v) for v in r[: len(group_by)]))
    if not group_by and not rows:
        rows = [[0 if a["agg"] in ("count", "count_star") else None for a in query["select"]]]
    return {"columns": group_by + aggcols, "rows": rows}
