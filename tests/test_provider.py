"""Metadata-plane tests of the provider (no GPU): manifest selection,
file-level min/max pruning (port of can_be_pruned/satisfy_constraints,
stream_schema_provider.rs:1049-1137 — cases re-derived from the semantics the
reference's own tests pin at :1360-1628), the count fast path
(query.rs:189-256), and the Final-merge of partial aggregate batches."""

import pyarrow as pa

from parseable_amd.provider import (
    ManifestCountResult,
    StandardTableProvider,
    _file_pruned,
    merge_partials,
)


def _entry(**stats_cols):
    cols = []
    for name, st in stats_cols.items():
        cols.append({"name": name, "stats": st, "uncompressed_size": 0, "compressed_size": 0})
    return {"file_path": "f", "num_rows": 10, "file_size": 1, "columns": cols}


INT_10_20 = {"Int": {"min": 10, "max": 20}}
STR_A_M = {"String": {"min": "aaa", "max": "mmm"}}


def P(col, op, **kw):
    return {"col": col, "op": op, **kw}


def test_int_pruning_matrix():
    e = _entry(x=INT_10_20)
    # eq inside range -> kept; outside -> pruned
    assert not _file_pruned(e, [P("x", "eq", lit=15)])
    assert _file_pruned(e, [P("x", "eq", lit=5)])
    assert _file_pruned(e, [P("x", "eq", lit=25)])
    # lt: satisfied iff min < v
    assert _file_pruned(e, [P("x", "lt", lit=10)])
    assert not _file_pruned(e, [P("x", "lt", lit=11)])
    # le: min <= v
    assert not _file_pruned(e, [P("x", "le", lit=10)])
    assert _file_pruned(e, [P("x", "le", lit=9)])
    # gt: max > v
    assert _file_pruned(e, [P("x", "gt", lit=20)])
    assert not _file_pruned(e, [P("x", "gt", lit=19)])
    # ge: max >= v
    assert not _file_pruned(e, [P("x", "ge", lit=20)])
    assert _file_pruned(e, [P("x", "ge", lit=21)])
    # ne / contains never prune (reference behavior)
    assert not _file_pruned(e, [P("x", "ne", lit=15)])


def test_string_pruning():
    e = _entry(s=STR_A_M)
    assert not _file_pruned(e, [P("s", "eq", lit="bbb")])
    assert _file_pruned(e, [P("s", "eq", lit="zzz")])
    # type mismatch: int literal vs string stats -> cannot prune
    assert not _file_pruned(e, [P("s", "eq", lit=7)])


def test_between_decomposes_to_bounds():
    e = _entry(ts=INT_10_20)
    assert _file_pruned(e, [P("ts", "between", lo=21, hi=30)])
    assert _file_pruned(e, [P("ts", "between", lo=0, hi=9)])
    assert not _file_pruned(e, [P("ts", "between", lo=18, hi=30)])
    # hi-exclusive (the injected time filter): [20,25) keeps, [21,25) prunes?
    assert not _file_pruned(e, [P("ts", "between", lo=0, hi=10, hi_exclusive=False)])
    assert _file_pruned(e, [P("ts", "between", lo=0, hi=10, hi_exclusive=True)])


def test_no_stats_never_pruned():
    e = _entry(x=None)
    assert not _file_pruned(e, [P("x", "eq", lit=1)])
    assert not _file_pruned(e, [P("y", "eq", lit=1)])  # unknown column


def test_count_fast_path_from_manifest(golden):
    import os

    stream_dir = os.path.join(
        os.path.dirname(os.path.abspath(__file__)), "golden", "data", "g_c1"
    )
    provider = StandardTableProvider(stream_dir, session=None)
    q = {"select": [{"agg": "count_star"}]}
    res = provider.scan(q)
    assert isinstance(res, ManifestCountResult)
    assert res.rows() == [[120_000]]  # manifest num_rows sum, never scans


def test_count_fast_path_declined_for_partial_range(golden):
    import os

    gdir = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden", "data", "g_c1")
    provider = StandardTableProvider(gdir, session=None)
    from tests.golden_queries import BASE

    q = {"select": [{"agg": "count_star"}], "time_range": [BASE, BASE + 30_000]}
    # partial range -> must go to the scan engine; without a GPU that raises
    import pytest

    from parseable_amd import GpuqError

    with pytest.raises(GpuqError):
        provider.scan(q)


def _partial_batch(keys, presence, aggv, aggc, key_name="level"):
    return pa.record_batch(
        {
            key_name: pa.array(keys, type=pa.string()),
            "__presence": pa.array(presence, type=pa.int64()),
            "agg0": pa.array(aggv, type=pa.int64()),
            "agg0_count": pa.array(aggc, type=pa.int64()),
        }
    )


def test_merge_partials_sum_and_minmax():
    q = {"select": [{"agg": "sum", "col": "latency"}], "group_by": ["level"]}
    b1 = _partial_batch(["INFO", "WARN"], [3, 1], [30, 7], [3, 1])
    b2 = _partial_batch(["INFO", "ERROR"], [2, 5], [12, 100], [2, 5])
    rows = merge_partials([b1, b2], q)
    assert rows == [["ERROR", 100], ["INFO", 42], ["WARN", 7]]

    qm = {"select": [{"agg": "max", "col": "latency"}], "group_by": ["level"]}
    rows = merge_partials([b1, b2], qm)
    assert rows == [["ERROR", 100], ["INFO", 30], ["WARN", 7]]

    qc = {"select": [{"agg": "count_star"}], "group_by": ["level"]}
    b1c = _partial_batch(["INFO"], [3], [3], [3])
    b2c = _partial_batch(["INFO"], [2], [2], [2])
    assert merge_partials([b1c, b2c], qc) == [["INFO", 5]]


def test_merge_partials_null_key_sorts_last():
    q = {"select": [{"agg": "count_star"}], "group_by": ["k"]}
    b = pa.record_batch(
        {
            "k": pa.array([None, "a"], type=pa.string()),
            "__presence": pa.array([2, 3], type=pa.int64()),
            "agg0": pa.array([2, 3], type=pa.int64()),
            "agg0_count": pa.array([2, 3], type=pa.int64()),
        }
    )
    assert merge_partials([b], q) == [["a", 3], [None, 2]]


def test_merge_partials_empty():
    q = {"select": [{"agg": "count_star"}, {"agg": "sum", "col": "x"}]}
    assert merge_partials([], q) == [[0, None]]


def test_parse_iso_ms_accepts_both_chrono_forms():
    # chrono serde emits '...%S.%fZ' normally but '...%SZ' when the
    # fractional part is zero: both must parse, matching
    # the native planner's catalog.cpp parse_iso_ms.
    from parseable_amd.provider import _parse_iso_ms

    assert _parse_iso_ms("2025-09-01T00:01:02.500Z") == 1756684862500
    assert _parse_iso_ms("2025-09-01T00:01:02Z") == 1756684862000


def test_staging_window_boundary_minute_truncated(tmp_path):
    # is_within_staging_window (stream_schema_provider.rs:936-958): boundary
    # = (now - 5min) truncated to the minute, compared with >= on the upper
    # bound.
    import json as _json

    sdir = tmp_path / "stream"
    sdir.mkdir()
    (sdir / "stream.json").write_text(_json.dumps(
        {"snapshot": {"manifest_list": []}}))
    staging = tmp_path / "staging"
    staging.mkdir()
    now_ms = 1_700_000_000_000 + 37_123  # mid-minute "now"
    p = StandardTableProvider(str(sdir), None, staging_dir=str(staging),
                              now_ms=now_ms)
    boundary = (now_ms - 5 * 60_000) // 60_000 * 60_000
    assert p._staging_touches((0, boundary))          # exactly on: touches
    assert p._staging_touches((0, boundary + 1))
    assert not p._staging_touches((0, boundary - 1))  # within-minute end: not


def test_supports_filters_pushdown_boundary():
    # stream_schema_provider.rs:759-777 + expr_in_boundary :960-976:
    # Exact only for minute-aligned >,>=,<,<= (or the injected BETWEEN)
    # on p_timestamp; everything else Inexact.
    from parseable_amd.provider import supports_filters_pushdown as sfp

    m = 60_000
    assert sfp([{"col": "p_timestamp", "op": "ge", "lit": 5 * m}]) == ["exact"]
    assert sfp([{"col": "p_timestamp", "op": "lt", "lit": 5 * m}]) == ["exact"]
    # not minute-aligned -> Inexact (second()/nanosecond() != 0)
    assert sfp([{"col": "p_timestamp", "op": "ge", "lit": 5 * m + 1}]) == ["inexact"]
    # eq is not in the boundary op set
    assert sfp([{"col": "p_timestamp", "op": "eq", "lit": 5 * m}]) == ["inexact"]
    # non-time columns always Inexact
    assert sfp([{"col": "level", "op": "eq", "lit": "INFO"},
                {"col": "latency", "op": "ge", "lit": 0}]) == ["inexact", "inexact"]
    # injected time range: exact iff both bounds minute-aligned
    assert sfp([{"col": "p_timestamp", "op": "between", "lo": 0, "hi": 2 * m,
                 "hi_exclusive": True}]) == ["exact"]
    assert sfp([{"col": "p_timestamp", "op": "between", "lo": 0, "hi": 2 * m + 5,
                 "hi_exclusive": True}]) == ["inexact"]


def test_zstd_manifest_roundtrip(tmp_path):
    """Manifests written zstd-compressed (catalog/manifest.rs:53-110: frame
    magic sniff, level 3) plan identically to plain-JSON manifests — in the
    Python planner AND the native catalog planner."""
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from datagen.gen import gen_stream

    a = gen_stream(str(tmp_path / "plain"), "s", "c1", rows=12_000,
                   rows_per_file=4_000, seed=9)
    b = gen_stream(str(tmp_path / "zstd"), "s", "c1", rows=12_000,
                   rows_per_file=4_000, seed=9, manifest_codec="zstd")
    with open(b["manifest_files"] and os.path.join(
            str(tmp_path / "zstd"), "s/date=2025-09-01/manifest.json"), "rb") as fh:
        assert fh.read(4) == b"\x28\xb5\x2f\xfd"  # really compressed

    q = {"select": [{"agg": "count_star"}]}
    pa_ = StandardTableProvider(a["stream_dir"], None)
    pb_ = StandardTableProvider(b["stream_dir"], None)
    ra = pa_.scan(dict(q))
    rb = pb_.scan(dict(q))
    assert isinstance(ra, ManifestCountResult) and isinstance(rb, ManifestCountResult)
    assert ra.rows() == rb.rows() == [[12_000]]

    # native planner (host-side; GPUQ_FAKE_DEVICE plans without a GPU)
    os.environ["GPUQ_FAKE_DEVICE"] = "1"
    try:
        from parseable_amd.provider import GpuSession, GpuExecutionPlan

        sess = GpuSession(device_mask=1)
        plan = GpuExecutionPlan(sess, None, dict(q), stream_dir=b["stream_dir"])
        assert plan.fast_count == 12_000
    finally:
        del os.environ["GPUQ_FAKE_DEVICE"]


def _make_legacy_stream(tmp_path):
    """A stream whose FIRST day-of-files predates every manifest: generate 4
    minutes, then rewrite the manifest/snapshot to cover only minutes 2-3 —
    minutes 0-1 become pre-manifest data reachable only via prefix listing
    (listing_table_builder.rs:46-118)."""
    import json as _json
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from datagen.gen import BASE_TS_MS, MINUTE_MS, gen_stream

    info = gen_stream(str(tmp_path), "s", "c1", rows=16_000, rows_per_file=4_000,
                      seed=21)
    mdir = os.path.join(str(tmp_path), "s", "date=2025-09-01")
    with open(os.path.join(mdir, "manifest.json")) as fh:
        man = _json.load(fh)
    kept = [f for f in man["files"]
            if "minute=02" in f["file_path"] or "minute=03" in f["file_path"]]
    assert len(kept) == 2
    with open(os.path.join(mdir, "manifest.json"), "w") as fh:
        _json.dump({"version": "v2", "files": kept}, fh)
    snap_path = os.path.join(info["stream_dir"], "stream.json")
    with open(snap_path) as fh:
        sj = _json.load(fh)
    item = sj["snapshot"]["manifest_list"][0]
    lo = BASE_TS_MS + 2 * MINUTE_MS
    from datetime import datetime, timezone

    item["time_lower_bound"] = datetime.fromtimestamp(
        lo / 1000, tz=timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%f") + "Z"
    with open(snap_path, "w") as fh:
        _json.dump(sj, fh)
    return info, BASE_TS_MS, MINUTE_MS


def test_legacy_listing_discovers_premanifest_files(tmp_path):
    from parseable_amd.provider import EmptyScanResult

    info, base, minute = _make_legacy_stream(tmp_path)
    prov = StandardTableProvider(info["stream_dir"], None)
    # range covering everything: 2 legacy files (listed) + 2 manifested
    q = {"select": [{"agg": "count_star"}],
         "time_range": [base, base + 4 * minute]}
    legacy = prov._legacy_files(q["time_range"])
    assert len(legacy) == 2
    assert all("minute=00" in p or "minute=01" in p for p in legacy)
    # range entirely inside the manifested window: no listing
    assert prov._legacy_files([base + 2 * minute, base + 4 * minute]) == []
    # range entirely pre-manifest: listing only, manifests all pruned
    assert len(prov._legacy_files([base, base + 2 * minute])) == 2


def test_legacy_listing_requires_time_range_when_no_manifests(tmp_path):
    import json as _json

    sdir = tmp_path / "stream"
    sdir.mkdir()
    (sdir / "stream.json").write_text(_json.dumps(
        {"snapshot": {"manifest_list": []}}))
    prov = StandardTableProvider(str(sdir), None)
    import pytest as _pytest

    from parseable_amd.provider import GpuqError

    with _pytest.raises(GpuqError):
        prov.scan({"select": [{"agg": "count_star"}]})
