"""The parity comparison gate itself (oracle/compare.py): the 1-ULP float
gate used by every GPU parity test, and the rtol gate used for
order-dependent cross-checks (Acero / the scalar C restatement)."""

import math

from oracle.compare import FLOAT_RTOL, _ulp_diff, rows_equal, values_equal


def test_ulp_mapping_monotonic_and_symmetric():
    assert _ulp_diff(1.0, 1.0) == 0
    assert _ulp_diff(1.0, math.nextafter(1.0, 2.0)) == 1
    assert _ulp_diff(-1.0, math.nextafter(-1.0, -2.0)) == 1
    assert _ulp_diff(-1.0, math.nextafter(-1.0, 0.0)) == 1
    assert _ulp_diff(0.0, -0.0) == 0          # signed zeros are equal
    assert _ulp_diff(0.0, 5e-324) == 1        # smallest subnormal adjacency
    assert _ulp_diff(-5e-324, 5e-324) == 2    # across zero
    assert _ulp_diff(1.0, 2.0) == 2**52


def test_default_gate_is_one_ulp():
    a = 0.1 + 0.2                      # 0.30000000000000004
    assert values_equal(a, 0.3)        # 1 ulp apart: passes
    two = math.nextafter(math.nextafter(0.3, 1), 1)
    assert not values_equal(two, 0.3)  # 2 ulp: fails
    # integers stay bit-exact regardless
    assert not values_equal(5, 6)
    assert values_equal(None, None)
    assert not values_equal(None, 0)


def test_rtol_gate_for_order_dependent_engines():
    a, b = 1e12, 1e12 * (1 + 5e-10)
    assert not values_equal(a, b)                       # far beyond 1 ulp
    assert values_equal(a, b, float_rtol=FLOAT_RTOL)    # inside 1e-9 rtol
    assert not values_equal(1.0, 1.01, float_rtol=FLOAT_RTOL)


def test_rows_equal_shapes():
    assert rows_equal([[1, "a"]], [[1, "a"]])
    assert not rows_equal([[1]], [[1], [2]])
    assert not rows_equal([[1, 2]], [[1]])


def test_oracle_f64_sum_is_correctly_rounded():
    """The oracle's f64 sums go through math.fsum — the same correctly
    rounded result the GPU's 256-bit superaccumulator produces — so the
    1-ULP gate is meaningful end to end."""
    import numpy as np
    import pyarrow as pa

    from oracle import query_oracle as qo

    rng = np.random.default_rng(11)
    vals = rng.random(40_000)
    tbl = pa.table({
        "p_timestamp": pa.array(np.arange(40_000, dtype=np.int64),
                                type=pa.timestamp("ms")),
        "x": pa.array(vals, type=pa.float64()),
    })
    q = {"select": [{"agg": "sum", "col": "x"}, {"agg": "avg", "col": "x"}]}
    r = qo.execute([], q, extra_tables=[tbl])["rows"]
    exact = math.fsum(vals)
    assert r[0][0] == exact
    assert r[0][1] == exact / len(vals)
