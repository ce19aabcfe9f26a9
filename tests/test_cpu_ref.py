"""Pin the scalar C full-path restatement (oracle/cpu_ref.c: thrift footer ->
LZ4_RAW -> PLAIN/RLE_DICT/DELTA_BP decode -> filter -> group-by) against the
committed golden vectors. This is the independent check that our
understanding of the parquet dialect — the same understanding the HIP
decoders implement — is correct, without pyarrow in the loop."""

import pytest

from oracle import cpu_ref_runner
from oracle.compare import FLOAT_RTOL, assert_rows_equal
from tests.golden_queries import GOLDEN_QUERIES


def _all_cases():
    # ext queries use features the scalar C restatement does not cover
    # (utf8/f64 min-max, f64 predicates) — pinned by the pyarrow oracle
    return [f"{fx}/{q}" for fx, qs in GOLDEN_QUERIES.items()
            for q, spec in qs if not spec.get("ext")]


@pytest.fixture(scope="session", autouse=True)
def built():
    cpu_ref_runner.build()


@pytest.mark.parametrize("case", _all_cases())
def test_cpu_ref_matches_golden(golden, case):
    fx = case.split("/")[0]
    entry = golden["answers"][case]
    files = golden["fixtures"][fx]["files"]
    r = cpu_ref_runner.execute(files, entry["query"])
    # the scalar C restatement's sums are compensated but order-
    # dependent: rtol gate (the 1-ULP gate is GPU vs oracle)
    assert_rows_equal(r["rows"], entry["result"]["rows"], case,
                      float_rtol=FLOAT_RTOL)
