"""ctypes bindings for libgpuq.so (the C-ABI of include/gpuq.h).

The library is the product's compute path: if it cannot be loaded or no GPU
is visible, callers MUST fail — there is no CPU fallback anywhere in
parseable_amd (the oracle is test infrastructure only)."""

from __future__ import annotations

import ctypes as C
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_DIR, "libgpuq.so")


class GpuqFile(C.Structure):
    _fields_ = [
        ("path", C.c_char_p),
        ("row_groups", C.POINTER(C.c_int32)),
        ("n_row_groups", C.c_int32),
    ]


class GpuqPred(C.Structure):
    _fields_ = [
        ("column", C.c_char_p),
        ("op", C.c_int32),
        ("lit_kind", C.c_int32),
        ("i64", C.c_int64 * 2),
        ("f64", C.c_double * 2),
        ("str", C.c_char_p),
        ("hi_exclusive", C.c_int32),
    ]


class GpuqAgg(C.Structure):
    _fields_ = [("op", C.c_int32), ("column", C.c_char_p)]


class GpuqMetrics(C.Structure):
    _fields_ = [
        ("rows_scanned", C.c_int64),
        ("rows_out", C.c_int64),
        ("bytes_scanned", C.c_int64),
        ("rowgroup_bytes_total", C.c_int64),
        ("hbm_bytes_est", C.c_int64),
        ("kernel_ns", C.c_int64),
        ("exec_ns", C.c_int64),
        ("load_ns", C.c_int64),
        ("decomp_ns", C.c_int64),
        ("cache_hit_bytes", C.c_int64),
    ]


OPS = {"eq": 0, "ne": 1, "lt": 2, "le": 3, "gt": 4, "ge": 5, "between": 6, "contains": 7}
AGGS = {"count_star": 0, "count": 1, "sum": 2, "min": 3, "max": 4}

_lib = None


def load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(LIB_PATH):
        raise RuntimeError(
            f"libgpuq.so not built at {LIB_PATH} — run __graft_entry__.build(). "
            "parseable_amd has no CPU fallback."
        )
    lib = C.CDLL(LIB_PATH)
    lib.gpuq_session_create.restype = C.c_void_p
    lib.gpuq_session_create.argtypes = [C.c_uint64]
    lib.gpuq_session_destroy.argtypes = [C.c_void_p]
    lib.gpuq_last_error.restype = C.c_char_p
    lib.gpuq_last_error.argtypes = [C.c_void_p]
    lib.gpuq_device_count.restype = C.c_int32
    lib.gpuq_plan_build_from_stream.restype = C.c_void_p
    lib.gpuq_plan_build_from_stream.argtypes = [
        C.c_void_p, C.c_char_p,
        C.POINTER(C.c_char_p), C.c_int32,
        C.POINTER(GpuqPred), C.c_int32,
        C.POINTER(C.c_char_p), C.c_int32,
        C.POINTER(GpuqAgg), C.c_int32,
        C.c_int64, C.POINTER(C.c_int64),
    ]
    lib.gpuq_plan_build.restype = C.c_void_p
    lib.gpuq_plan_build.argtypes = [
        C.c_void_p,
        C.POINTER(GpuqFile), C.c_int32,
        C.POINTER(C.c_char_p), C.c_int32,
        C.POINTER(GpuqPred), C.c_int32,
        C.POINTER(C.c_char_p), C.c_int32,
        C.POINTER(GpuqAgg), C.c_int32,
        C.c_int64,
    ]
    lib.gpuq_plan_partition_count.restype = C.c_int32
    lib.gpuq_plan_partition_count.argtypes = [C.c_void_p]
    lib.gpuq_plan_load.restype = C.c_int32
    lib.gpuq_plan_load.argtypes = [C.c_void_p, C.c_int32]
    lib.gpuq_plan_execute.restype = C.c_int32
    lib.gpuq_plan_execute.argtypes = [C.c_void_p, C.c_int32, C.c_void_p]
    lib.gpuq_plan_metrics.restype = C.c_int32
    lib.gpuq_plan_metrics.argtypes = [C.c_void_p, C.POINTER(GpuqMetrics)]
    lib.gpuq_plan_destroy.argtypes = [C.c_void_p]
    _lib = lib
    return lib


def device_count() -> int:
    return load().gpuq_device_count()
