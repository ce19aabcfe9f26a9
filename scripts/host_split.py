#!/usr/bin/env python3
"""Diagnose the wall-vs-kernel gap per step: time plan.execute (C side:
event sync + Arrow export) vs merger.step (python Final merge) separately,
and report the library's own exec_ns/kernel_ns metrics deltas."""
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from datagen.gen import gen_stream  # noqa: E402
from parseable_amd import GpuSession, StandardTableProvider  # noqa: E402
from parseable_amd.dist import DistMerger  # noqa: E402
from bench import WORKLOADS  # noqa: E402


def main():
    wl = sys.argv[1] if len(sys.argv) > 1 else "c2s"
    rows = int(sys.argv[2]) if len(sys.argv) > 2 else 100_000_000
    cfg, qb = WORKLOADS[wl]

    class A:  # bench arg defaults
        between_ms = 150 * 60_000

    query = qb(A)
    shard = os.path.join("/tmp/gpuq_bench", f"{wl}_{rows}")
    if not os.path.exists(os.path.join(shard, "stream", "stream.json")):
        gen_stream(shard, "stream", cfg, rows=rows, seed=42,
                   workers=os.cpu_count() or 8)
    sess = GpuSession(device_mask=1)
    prov = StandardTableProvider(os.path.join(shard, "stream"), sess)
    plan = prov.scan(query)
    plan.load()
    merger = DistMerger(query, device="cpu")
    merger.setup(plan.execute(0))
    for _ in range(3):
        merger.step(plan.execute(0))
    m0 = plan.metrics()
    te = tm = 0.0
    N = 10
    t_all = time.perf_counter()
    for _ in range(N):
        t0 = time.perf_counter()
        b = plan.execute(0)
        t1 = time.perf_counter()
        merger.step(b)
        t2 = time.perf_counter()
        te += t1 - t0
        tm += t2 - t1
    t_all = time.perf_counter() - t_all
    m1 = plan.metrics()
    print(f"{wl}: step={t_all / N * 1e3:.3f}ms  plan.execute={te / N * 1e3:.3f}ms "
          f"merger.step={tm / N * 1e3:.3f}ms")
    print(f"  lib: kernel={(m1['kernel_ns'] - m0['kernel_ns']) / N / 1e6:.3f}ms "
          f"exec={(m1['exec_ns'] - m0['exec_ns']) / N / 1e6:.3f}ms")
    plan.close()


if __name__ == "__main__":
    main()
