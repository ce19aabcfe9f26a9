"""Result comparison for parity tests.

Integer, count, byte and key results must be BIT-EXACT (BASELINE.json gate:
"bit-exact counts/row sets"). Floating-point sums are order-dependent; the
reference engine (DataFusion partial->final agg) itself produces different
roundings per partitioning, so float aggregates compare within FLOAT_RTOL
(documented here, asserted in tests): |a-b| <= FLOAT_RTOL * max(|a|,|b|) + 1e-300.
FLOAT_RTOL = 1e-9 ≈ n·eps headroom for n ≈ 1e7-row group sums; min/max on
floats remain bit-exact (order-independent).
"""

FLOAT_RTOL = 1e-9


def values_equal(a, b, float_rtol=FLOAT_RTOL):
    if a is None or b is None:
        return a is None and b is None
    if isinstance(a, float) or isinstance(b, float):
        fa, fb = float(a), float(b)
        if fa == fb:
            return True
        return abs(fa - fb) <= float_rtol * max(abs(fa), abs(fb)) + 1e-300
    return a == b


def rows_equal(rows1, rows2, float_rtol=FLOAT_RTOL):
    if len(rows1) != len(rows2):
        return False
    for r1, r2 in zip(rows1, rows2):
        if len(r1) != len(r2):
            return False
        for a, b in zip(r1, r2):
            if not values_equal(a, b, float_rtol):
                return False
    return True


def assert_rows_equal(rows1, rows2, msg="", float_rtol=FLOAT_RTOL):
    assert rows_equal(rows1, rows2, float_rtol), (
        f"{msg}\nlhs={rows1[:20]}\nrhs={rows2[:20]}"
        + (f"\n({len(rows1)} vs {len(rows2)} rows)" if len(rows1) != len(rows2) else "")
    )
