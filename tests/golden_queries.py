"""Golden query suite over the committed fixtures (tests/golden/data).

Query IR is the plan-level surface of SURVEY.md §8b: conjunctive predicates
(col op literal | BETWEEN | LIKE-'%x%'), optional GROUP BY keys, aggregate
list, plus the injected time range [start, end) in ms
(src/query/mod.rs:829-888). Edge cases mirror what the reference's own tests
exercise at planning level (stream_schema_provider.rs:1139-1629) plus
value-level cases the reference leaves to its engine: empty results,
all-match, NULL-heavy columns (c4), dict→PLAIN fallback pages (c1 latency,
c3 message), multi-file merges, single-row files.
"""

# time base must match datagen.gen.BASE_TS_MS
BASE = 1756684800000  # 2025-09-01T00:00:00Z in ms
MIN = 60_000

# fixture name -> list of (query_name, query_dict)
GOLDEN_QUERIES = {
    "g_c1": [
        ("count_by_level", {
            "select": [{"agg": "count_star"}],
            "group_by": ["level"],
        }),
        ("count_max_by_host_between", {
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["host"],
            "preds": [{"col": "p_timestamp", "op": "between",
                       "lo": BASE + MIN // 2, "hi": BASE + 2 * MIN}],
        }),
        ("sum_min_where_host_eq", {
            "select": [{"agg": "sum", "col": "latency"},
                       {"agg": "min", "col": "latency"},
                       {"agg": "count_star"}],
            "preds": [{"col": "host", "op": "eq", "lit": "host-0001"}],
        }),
        ("empty_result", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
            "preds": [{"col": "host", "op": "eq", "lit": "no-such-host"}],
        }),
        ("all_match", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
            "preds": [{"col": "latency", "op": "ge", "lit": 0}],
        }),
        ("time_range_injected", {
            "select": [{"agg": "count_star"}],
            "group_by": ["level"],
            "time_range": [BASE + MIN, BASE + 2 * MIN],
        }),
        ("lt_on_i64", {
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "f_i64"},
                       {"agg": "min", "col": "f_i64"}],
            "preds": [{"col": "latency", "op": "lt", "lit": 1000}],
        }),
        ("ne_and_two_keys", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
            "group_by": ["level", "f_str2"],
            "preds": [{"col": "f_str1", "op": "ne", "lit": "region-0"}],
        }),
        ("sum_f64", {
            "select": [{"agg": "sum", "col": "f_f64"}, {"agg": "count_star"}],
            "group_by": ["level"],
            "preds": [{"col": "level", "op": "ge", "lit": "INFO"}],
        }),
        ("avg_latency_by_level", {
            "select": [{"agg": "avg", "col": "latency"}, {"agg": "count_star"}],
            "group_by": ["level"],
        }),
        ("avg_f64_by_key", {
            "select": [{"agg": "avg", "col": "f_f64"},
                       {"agg": "sum", "col": "f_f64"}],
            "group_by": ["f_str2"],
            "preds": [{"col": "latency", "op": "ge", "lit": 250_000}],
        }),
        # extended coverage (ext: skipped by the scalar C oracle — pinned by
        # the pyarrow oracle + Acero instead)
        ("minmax_utf8", {
            "ext": True,
            "select": [{"agg": "min", "col": "host"}, {"agg": "max", "col": "host"},
                       {"agg": "count_star"}],
            "group_by": ["level"],
        }),
        ("minmax_f64", {
            "ext": True,
            "select": [{"agg": "min", "col": "f_f64"}, {"agg": "max", "col": "f_f64"}],
            "group_by": ["f_str2"],
        }),
        ("pred_f64", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
            "preds": [{"col": "f_f64", "op": "lt", "lit": 0.25}],
        }),
        # fuzz-found regressions: a utf8 column serving several roles at
        # once (group key / min-max rank agg / count / predicate) — aux-slot
        # aliasing produced garbage gids and OOB group-table writes
        ("key_and_minmax_same_col", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "f_str2"}],
            "group_by": ["host", "f_str2"],
        }),
        ("rank_count_other_col", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "f_str1"},
                       {"agg": "count", "col": "f_str1"}],
            "group_by": ["f_str2", "level"],
        }),
        # numeric group keys (DataFusion groups by any column; gids from
        # the value hash — pinned by pyarrow oracle + Acero, the scalar C
        # restatement keys strings only)
        ("int_key_group", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "f_i64"}],
            "group_by": ["latency"],
            "preds": [{"col": "latency", "op": "lt", "lit": 2000}],
        }),
        ("int_key_mixed_pair", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "min", "col": "f_f64"}],
            "group_by": ["level", "latency"],
            "preds": [{"col": "latency", "op": "lt", "lit": 800}],
        }),
        ("ts_minmax", {
            # time span per group: min/max over the timestamp column
            # (i64 ms; DELTA_BINARY_PACKED pages)
            "select": [{"agg": "min", "col": "p_timestamp"},
                       {"agg": "max", "col": "p_timestamp"},
                       {"agg": "count_star"}],
            "group_by": ["level"],
            "preds": [{"col": "host", "op": "eq", "lit": "host-0007"}],
        }),
        ("f64_key_group", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["f_f64"],
            "preds": [{"col": "f_f64", "op": "lt", "lit": 0.002}],
        }),
        ("ts_key_group", {
            "ext": True,
            "select": [{"agg": "count_star"}],
            "group_by": ["p_timestamp"],
            "time_range": [BASE, BASE + 2_000],
        }),
        ("key_rank_pred_same_col", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "min", "col": "level"}],
            "group_by": ["host", "level"],
            "preds": [{"col": "level", "op": "ne", "lit": "DEBUG"}],
        }),
    ],
    "g_c1_pages": [
        ("count_by_level", {
            "select": [{"agg": "count_star"}],
            "group_by": ["level"],
        }),
        ("between_half", {
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["host"],
            "preds": [{"col": "p_timestamp", "op": "between",
                       "lo": BASE, "hi": BASE + MIN // 2}],
        }),
    ],
    "g_c3": [
        ("like_error", {
            "select": [{"agg": "count_star"}],
            "preds": [{"col": "message", "op": "contains", "lit": "error"}],
        }),
        ("like_error_by_level", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
            "group_by": ["level"],
            "preds": [{"col": "message", "op": "contains", "lit": "error"}],
        }),
        ("like_no_match", {
            "select": [{"agg": "count_star"}],
            "preds": [{"col": "message", "op": "contains", "lit": "zzzzzzzzzzzz"}],
        }),
        ("like_two_needles", {
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["level"],
            "preds": [{"col": "message", "op": "contains", "lit": "err"},
                      {"col": "message", "op": "contains", "lit": "ror"}],
        }),
    ],
    "g_c4": [
        ("three_key_groupby", {
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["service", "span_kind", "status"],
        }),
        ("count_nullable_col", {
            "select": [{"agg": "count_star"}, {"agg": "count", "col": "attr_s0"},
                       {"agg": "count", "col": "attr_i5"}],
        }),
        ("group_by_nullable", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "attr_i0"}],
            "group_by": ["attr_s1"],
        }),
        ("int_key_nullable", {
            "ext": True,
            "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
            "group_by": ["attr_i5"],
            "preds": [{"col": "attr_i0", "op": "lt", "lit": 30000}],
        }),
        ("pred_on_nullable", {
            "select": [{"agg": "count_star"}],
            "preds": [{"col": "attr_i1", "op": "ge", "lit": 500000}],
            "group_by": ["status"],
        }),
    ],
    "g_edge": [
        ("tiny_files_count", {
            "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"},
                       {"agg": "min", "col": "latency"}, {"agg": "max", "col": "latency"}],
            "group_by": ["level"],
        }),
        ("tiny_between_none", {
            "select": [{"agg": "count_star"}],
            "time_range": [0, 1],
        }),
    ],
}

# fixture name -> datagen parameters
GOLDEN_QUERIES["g_c5"] = [
    ("hash_group_count", {
        "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
        "group_by": ["trace"],
    }),
    ("hash_two_keys_mixed", {
        # hash key (trace, PLAIN fallback) x dict key (level, pure dict)
        "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
        "group_by": ["level", "trace"],
        "preds": [{"col": "latency", "op": "ge", "lit": 100_000}],
    }),
    ("hash_minmax_str", {
        "ext": True,  # utf8 min/max: pinned by pyarrow oracle + Acero
        "select": [{"agg": "min", "col": "trace"}, {"agg": "max", "col": "trace"},
                   {"agg": "count_star"}],
        "group_by": ["level"],
    }),
    ("hash_eq_pred_plain", {
        # non-LIKE predicate on a PLAIN-fallback utf8 column
        "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
        "preds": [{"col": "trace", "op": "eq", "lit": "tr-00000042"}],
    }),
    ("hash_nullable_key", {
        # nullable hash key: NULL group + def-level routing on PLAIN pages
        "select": [{"agg": "count_star"}],
        "group_by": ["opt_tag"],
    }),
    ("hash_count_nullable", {
        "select": [{"agg": "count", "col": "opt_tag"}, {"agg": "count_star"}],
        "group_by": ["level"],
    }),
    ("hash_contains_on_key", {
        # contains predicate riding the strref path (hash mode via group key)
        "select": [{"agg": "count_star"}],
        "group_by": ["trace"],
        "preds": [{"col": "trace", "op": "contains", "lit": "42"}],
    }),
    ("hash_ge_pred_str", {
        "select": [{"agg": "count_star"}],
        "preds": [{"col": "opt_tag", "op": "ge", "lit": "tag-000600"}],
    }),
    ("hash_cascade_three_keys", {
        # dense key-product space 1501x1001x801 >> the 2^22 cap: the exact
        # pair cascade combines keys two at a time on device
        "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
        "group_by": ["trace", "host", "opt_tag"],
        "preds": [{"col": "latency", "op": "lt", "lit": 15_000}],
    }),
    ("hash_avg_by_tag", {
        "select": [{"agg": "avg", "col": "latency"}, {"agg": "count_star"}],
        "group_by": ["opt_tag"],
        "time_range": [BASE, BASE + 2 * MIN],
    }),
]

GOLDEN_QUERIES["g_c1snap"] = [
    ("snap_count_by_level", {
        "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
        "group_by": ["level"],
    }),
    ("snap_between_pred", {
        "select": [{"agg": "count_star"}, {"agg": "sum", "col": "latency"}],
        "group_by": ["host"],
        "preds": [{"col": "p_timestamp", "op": "between",
                   "lo": BASE + MIN // 2, "hi": BASE + 2 * MIN}],
    }),
    ("snap_sum_f64", {
        "select": [{"agg": "sum", "col": "f_f64"}, {"agg": "count_star"}],
        "group_by": ["f_str2"],
        "preds": [{"col": "level", "op": "eq", "lit": "INFO"}],
    }),
]

GOLDEN_FIXTURES = {
    "g_c1":       dict(config="c1", rows=120_000, rows_per_file=40_000, seed=1001),
    "g_c1_pages": dict(config="c1", rows=40_000, rows_per_file=40_000, seed=1002,
                       data_page_size=8 * 1024),
    "g_c3":       dict(config="c3", rows=25_000, rows_per_file=25_000, seed=1003),
    "g_c4":       dict(config="c4", rows=20_000, rows_per_file=20_000, seed=1004),
    "g_edge":     dict(config="c1", rows=9, rows_per_file=4, seed=1005),
    # raw-byte utf8: dict-overflow -> PLAIN-fallback pages (written with a
    # tiny dictionary_pagesize_limit); exercises the device hash group-by,
    # strref predicates and strref min/max
    "g_c5":       dict(config="c5", rows=60_000, rows_per_file=20_000, seed=1006),
    # snappy page codec (the reference's test-compose codec,
    # docker-compose-test.yaml:45) + zstd-compressed manifests
    # (catalog/manifest.rs:53-110)
    "g_c1snap":   dict(config="c1", rows=60_000, rows_per_file=20_000,
                       seed=1007, compression="snappy", manifest_codec="zstd"),
}
