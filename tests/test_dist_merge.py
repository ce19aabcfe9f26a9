"""Multi-process (gloo, world_size=2) test of the cross-rank Final merge —
the exact path bench.py uses for the RCCL merge on an 8-GPU node
(SURVEY.md §8e), run here on CPU with the gloo backend."""

import os

import pyarrow as pa
import pytest
import torch.multiprocessing as mp


def _partial(keys, presence, vals, cnts):
    return pa.record_batch(
        {
            "level": pa.array(keys, type=pa.string()),
            "__presence": pa.array(presence, type=pa.int64()),
            "agg0": pa.array([p for p in presence], type=pa.int64()),
            "agg0_count": pa.array([p for p in presence], type=pa.int64()),
            "agg1": pa.array(vals, type=pa.int64()),
            "agg1_count": pa.array(cnts, type=pa.int64()),
        }
    )


def _partial2(keys, presence, aggs):
    """aggs: list of (values, counts, pa type) per aggregate slot."""
    cols = {"level": pa.array(keys, type=pa.string()),
            "__presence": pa.array(presence, type=pa.int64())}
    for i, (vals, cnts, typ) in enumerate(aggs):
        cols[f"agg{i}"] = pa.array(vals, type=typ)
        cols[f"agg{i}_count"] = pa.array(cnts, type=pa.int64())
    return pa.record_batch(cols)


QUERY = {
    "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
    "group_by": ["level"],
}

EXPECTED = [["ERROR", 5, 400], ["INFO", 5, 300], ["WARN", 4, 77]]

# float64 min/max across ranks, with rank 1 contributing NOTHING (batch
# None): the agreed per-agg types must come from the gather, not the local
# batch (ADVICE round 1, high + medium findings).
QUERY_F = {
    "select": [{"agg": "min", "col": "f_f64"}, {"agg": "max", "col": "f_f64"}],
    "group_by": ["level"],
}
EXPECTED_F = [["INFO", 1.5, 9.0], ["WARN", 2.5, 7.25]]

# utf8 min/max across ranks (was NotImplementedError)
QUERY_S = {
    "select": [{"agg": "min", "col": "host"}, {"agg": "max", "col": "host"}],
    "group_by": ["level"],
}
EXPECTED_S = [["INFO", "a-host", "z-host"], ["WARN", "b", "b"]]


def _rank_main(rank, world, port, q):
    import torch.distributed as dist

    from parseable_amd.dist import DistMerger

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    try:
        out = {}
        if rank == 0:
            batch = _partial(["INFO", "WARN"], [3, 4], [300, 77], [3, 4])
        else:
            batch = _partial(["INFO", "ERROR"], [2, 5], [250, 400], [2, 5])
        m = DistMerger(QUERY, device="cpu")
        m.setup(batch)
        out["int"] = m.step(batch)

        if rank == 0:
            fb = _partial2(["INFO", "WARN"], [3, 4],
                           [([1.5, 2.5], [3, 4], pa.float64()),
                            ([9.0, 7.25], [3, 4], pa.float64())])
        else:
            fb = None
        m = DistMerger(QUERY_F, device="cpu")
        m.setup(fb)
        out["f64"] = m.step(fb)

        if rank == 0:
            sb = _partial2(["INFO"], [3],
                           [(["a-host"], [3], pa.string()),
                            (["m-host"], [3], pa.string())])
        else:
            sb = _partial2(["INFO", "WARN"], [2, 1],
                           [(["c", "b"], [2, 1], pa.string()),
                            (["z-host", "b"], [2, 1], pa.string())])
        m = DistMerger(QUERY_S, device="cpu")
        m.setup(sb)
        out["str"] = m.step(sb)
        q.put((rank, out))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gloo_world2_merge():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, rows = q.get()
        results[rank] = rows
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    for r in (0, 1):   # every rank sees the same final table
        assert results[r]["int"] == EXPECTED
        assert results[r]["f64"] == EXPECTED_F
        assert results[r]["str"] == EXPECTED_S


def _rank_main4(rank, world, port, q):
    import torch.distributed as dist

    from parseable_amd.dist import DistMerger

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    try:
        # world-4: mixed avg/f64-sum/utf8-max slots, one empty rank, disjoint
        # and overlapping key spaces
        qm = {"select": [{"agg": "count_star"}, {"agg": "avg", "col": "latency"},
                         {"agg": "sum", "col": "f_f64"},
                         {"agg": "max", "col": "host"}],
              "group_by": ["level"]}
        batches = [
            _partial2(["INFO", "WARN"], [2, 1],
                      [([2, 1], [2, 1], pa.int64()),     # count_star mirror
                       ([10, 7], [2, 1], pa.int64()),    # avg: sums
                       ([0.5, 0.25], [2, 1], pa.float64()),
                       (["h-b", "h-a"], [2, 1], pa.string())]),
            _partial2(["INFO"], [3],
                      [([3], [3], pa.int64()),
                       ([30], [3], pa.int64()),
                       ([1.5], [3], pa.float64()),
                       (["h-z"], [3], pa.string())]),
            None,
            _partial2(["ERROR"], [4],
                      [([4], [4], pa.int64()),
                       ([100], [4], pa.int64()),
                       ([2.0], [4], pa.float64()),
                       (["h-m"], [4], pa.string())]),
        ]
        m = DistMerger(qm, device="cpu")
        m.setup(batches[rank])
        rows = m.step(batches[rank])
        q.put((rank, rows))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_gloo_world4_mixed_aggs():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_rank_main4, args=(r, 4, port, q)) for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, rows = q.get()
        results[rank] = rows
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    expected = [
        ["ERROR", 4, 100 / 4, 2.0, "h-m"],
        ["INFO", 5, 40 / 5, 2.0, "h-z"],
        ["WARN", 1, 7 / 1, 0.25, "h-a"],
    ]
    for r in range(4):
        assert results[r] == expected
