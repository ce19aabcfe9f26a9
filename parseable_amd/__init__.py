"""parseable_amd — MI355X-native query-execution path for Parseable log
streams (the DataFusion physical-plan replacement of SURVEY.md §8).

Compute runs in libgpuq.so (hand-written HIP for gfx950 behind the C ABI of
include/gpuq.h); this package is the host-side mirror of the reference's
TableProvider/ExecutionPlan surface plus the metadata plumbing (manifest
JSON, pruning, partial-aggregate merge). There is NO CPU fallback: on a
machine without a GPU every compute entry point raises."""

from .provider import (  # noqa: F401
    EmptyScanResult,
    GpuSession,
    GpuExecutionPlan,
    GpuqError,
    ManifestCountResult,
    Query,
    StandardTableProvider,
    merge_partials,
)
from .provider import merge_topk  # noqa: F401

__version__ = "0.1.0"
