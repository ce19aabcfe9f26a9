"""Result comparison for parity tests.

Integer, count, byte and key results must be BIT-EXACT (BASELINE.json gate:
"bit-exact counts/row sets").

Floating-point aggregates (the default gate, used by every GPU parity test
and the fuzzer): within 1 ULP of the oracle. This honors BASELINE.json's
"sum/avg within 1 ULP" literally because BOTH sides now produce the
correctly rounded exact sum — the oracle via math.fsum, the GPU via the
256-bit fixed-point superaccumulator (kernels.hip acc256_*, rounded once at
export) — so they normally agree to 0 ULP; min/max on floats are
order-independent and bit-exact.

FLOAT_RTOL is for cross-checks against engines whose sums are
order-DEPENDENT (pyarrow Acero, the scalar C restatement's compensated
sums): |a-b| <= rtol * max(|a|,|b|) + 1e-300. Pass float_rtol=FLOAT_RTOL
explicitly for those; it is never the GPU parity gate.
"""

import struct

FLOAT_RTOL = 1e-9


def _ulp_diff(a: float, b: float) -> int:
    """Distance in representable doubles, via the monotonic integer mapping
    of IEEE-754 bit patterns (negatives reflected so adjacency is uniform;
    +0.0 and -0.0 both map to 0)."""
    ia = struct.unpack("<q", struct.pack("<d", a))[0]
    ib = struct.unpack("<q", struct.pack("<d", b))[0]
    if ia < 0:
        ia = -(2**63) - ia
    if ib < 0:
        ib = -(2**63) - ib
    return abs(ia - ib)


def values_equal(a, b, float_rtol=None, float_ulps=1):
    if a is None or b is None:
        return a is None and b is None
    if isinstance(a, float) or isinstance(b, float):
        fa, fb = float(a), float(b)
        if fa == fb:
            return True
        if float_rtol is not None:
            return abs(fa - fb) <= float_rtol * max(abs(fa), abs(fb)) + 1e-300
        return _ulp_diff(fa, fb) <= float_ulps
    return a == b


def rows_equal(rows1, rows2, float_rtol=None, float_ulps=1):
    if len(rows1) != len(rows2):
        return False
    for r1, r2 in zip(rows1, rows2):
        if len(r1) != len(r2):
            return False
        for a, b in zip(r1, r2):
            if not values_equal(a, b, float_rtol, float_ulps):
                return False
    return True


def assert_rows_equal(rows1, rows2, msg="", float_rtol=None, float_ulps=1):
    assert rows_equal(rows1, rows2, float_rtol, float_ulps), (
        f"{msg}\nlhs={rows1[:20]}\nrhs={rows2[:20]}"
        + (f"\n({len(rows1)} vs {len(rows2)} rows)" if len(rows1) != len(rows2) else "")
    )
