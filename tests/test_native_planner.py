"""The native catalog planner (catalog.cpp: stream.json/manifest parse,
manifest selection, min/max pruning, count fast path) must make the SAME
planning decisions as the Python mirror (provider.py) whose semantics are
pinned against the reference in test_provider.py. Runs on CPU via
GPUQ_FAKE_DEVICE (plan building is pure host work)."""

import os

import pytest

from tests.golden_queries import GOLDEN_QUERIES

GDIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden", "data")


@pytest.fixture(scope="module", autouse=True)
def fake_device(request):
    os.environ["GPUQ_FAKE_DEVICE"] = "1"
    yield
    os.environ.pop("GPUQ_FAKE_DEVICE", None)


def _plan_stats(provider, query, native: bool):
    from parseable_amd import (EmptyScanResult, GpuExecutionPlan,
                               ManifestCountResult)

    if native:
        os.environ.pop("GPUQ_PY_PLANNER", None)
    else:
        os.environ["GPUQ_PY_PLANNER"] = "1"
    try:
        plan = provider.scan(query)
    finally:
        os.environ.pop("GPUQ_PY_PLANNER", None)
    if isinstance(plan, ManifestCountResult):
        return ("count", plan.count)
    if isinstance(plan, EmptyScanResult):
        return ("empty",)
    try:
        m = plan.metrics()
        return ("scan", m["rows_scanned"], m["bytes_scanned"],
                m["rowgroup_bytes_total"])
    finally:
        plan.close()


def _all_cases():
    return [f"{fx}/{q}" for fx, qs in GOLDEN_QUERIES.items() for q, _ in qs]


@pytest.mark.parametrize("case", _all_cases())
def test_native_planner_matches_python(golden, case):
    from parseable_amd import GpuSession, StandardTableProvider

    fx = case.split("/")[0]
    query = golden["answers"][case]["query"]
    provider = StandardTableProvider(os.path.join(GDIR, fx), GpuSession())
    a = _plan_stats(provider, query, native=True)
    b = _plan_stats(provider, query, native=False)
    assert a == b, f"{case}: native={a} python={b}"
