"""Pin the numpy oracle against (a) the committed golden vectors and (b)
pyarrow Acero re-executed live — the two independent engines must agree on
every golden query (parity anchored on our own harness since
the reference's Rust engine cannot compile here)."""

import pytest

from oracle import query_oracle as qo
from oracle.compare import FLOAT_RTOL, assert_rows_equal
from tests.golden_queries import GOLDEN_QUERIES


def _all_cases():
    out = []
    for fx, qs in GOLDEN_QUERIES.items():
        for qname, _ in qs:
            out.append(f"{fx}/{qname}")
    return out


@pytest.mark.parametrize("case", _all_cases())
def test_oracle_matches_golden(golden, case):
    fx = case.split("/")[0]
    entry = golden["answers"][case]
    files = golden["fixtures"][fx]["files"]
    r = qo.execute(files, entry["query"])
    assert_rows_equal(r["rows"], entry["result"]["rows"], case)


@pytest.mark.parametrize("case", _all_cases())
def test_acero_matches_golden(golden, case):
    fx = case.split("/")[0]
    entry = golden["answers"][case]
    files = golden["fixtures"][fx]["files"]
    r = qo.execute_acero(files, entry["query"])
    # Acero's float sums are order-dependent: rtol gate, not the 1-ULP gate
    assert_rows_equal(r["rows"], entry["result"]["rows"], case,
                      float_rtol=FLOAT_RTOL)


def test_dialect_of_golden_files(golden):
    """The committed fixtures really are Parseable's parquet dialect
    (src/cli.rs:468-491, src/parseable/streams.rs:705-780)."""
    import pyarrow.parquet as pq

    files = golden["fixtures"]["g_c1"]["files"]
    md = pq.read_metadata(files[0])
    assert md.num_rows <= 262_144
    rg = md.row_group(0)
    names = [rg.column(i).path_in_schema for i in range(rg.num_columns)]
    ts_i = names.index("p_timestamp")
    c = rg.column(ts_i)
    assert "DELTA_BINARY_PACKED" in c.encodings
    assert c.compression == "LZ4"  # pyarrow's display name for codec 7 LZ4_RAW
    # time-DESC sort advertised
    assert md.row_group(0).sorting_columns[0].descending
    lvl = rg.column(names.index("level"))
    assert "RLE_DICTIONARY" in lvl.encodings


def test_oracle_date_bin_semantics(golden):
    """DATE_BIN keys: origin-aligned windows; counts complete."""
    from tests.golden_queries import BASE

    files = golden["fixtures"]["g_c1"]["files"]
    q = {"select": [{"agg": "count_star"}],
         "group_by": [{"bin": "p_timestamp", "stride_ms": 60_000, "origin": 0}]}
    r = qo.execute(files, q)
    assert sum(row[1] for row in r["rows"]) == 120_000
    for row in r["rows"]:
        assert row[0] % 60_000 == 0
        assert BASE <= row[0] < BASE + 10 * 60_000
    # window widths partition rows: 2-minute bins merge adjacent 1-minute bins
    q2 = {"select": [{"agg": "count_star"}],
          "group_by": [{"bin": "p_timestamp", "stride_ms": 120_000, "origin": BASE}]}
    r2 = qo.execute(files, q2)
    ones = {row[0]: row[1] for row in r["rows"]}
    for row in r2["rows"]:
        assert row[1] == ones.get(row[0], 0) + ones.get(row[0] + 60_000, 0)
