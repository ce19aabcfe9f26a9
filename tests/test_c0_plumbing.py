"""BASELINE configs[0]: the 10M-row demo-stream shape
(resources/ingest_demo_data.sh fields), SELECT count(*) WHERE host='x' on
the CPU path — a plumbing check with no GPU: the c0 dialect parses, the
planner's manifest selection/pruning behaves, and all three CPU
restatements agree on the c0 query. (Scaled down for the suite; the shape
matches the config.)"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest

from datagen.gen import BASE_TS_MS, MINUTE_MS, gen_stream
from oracle import query_oracle as qo
from oracle.compare import FLOAT_RTOL, assert_rows_equal
from parseable_amd.provider import ManifestCountResult, StandardTableProvider

Q = {"select": [{"agg": "count_star"}],
     "preds": [{"col": "host", "op": "eq", "lit": "host-0003"}]}


@pytest.fixture(scope="module")
def c0_stream(tmp_path_factory):
    td = tmp_path_factory.mktemp("c0")
    return gen_stream(str(td), "demo", "c0", rows=60_000, rows_per_file=20_000,
                      seed=3003)


def test_c0_oracles_agree(c0_stream):
    r1 = qo.execute(c0_stream["files"], dict(Q))
    r2 = qo.execute_acero(c0_stream["files"], dict(Q))
    assert_rows_equal(r1["rows"], r2["rows"], "c0 acero", float_rtol=FLOAT_RTOL)
    assert r1["rows"][0][0] > 0
    from oracle import cpu_ref_runner

    r3 = cpu_ref_runner.execute(c0_stream["files"], dict(Q))
    assert_rows_equal(r1["rows"], r3["rows"], "c0 scalar C",
                      float_rtol=FLOAT_RTOL)


def test_c0_planner_fast_count_and_pruning(c0_stream):
    prov = StandardTableProvider(c0_stream["stream_dir"], None)
    # bare count: answered from manifest sums, never reaches a scan
    plan = prov.scan({"select": [{"agg": "count_star"}]})
    assert isinstance(plan, ManifestCountResult)
    assert plan.rows() == [[60_000]]
    # minute-level time range keeps exactly one file's rows
    plan2 = prov.scan({"select": [{"agg": "count_star"}],
                       "time_range": [BASE_TS_MS + MINUTE_MS,
                                      BASE_TS_MS + 2 * MINUTE_MS]})
    assert isinstance(plan2, ManifestCountResult)
    assert plan2.rows() == [[20_000]]
