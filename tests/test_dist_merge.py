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


QUERY = {
    "select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
    "group_by": ["level"],
}

EXPECTED = [["ERROR", 5, 400], ["INFO", 5, 300], ["WARN", 4, 77]]


def _rank_main(rank, world, port, q):
    import torch.distributed as dist

    from parseable_amd.dist import DistMerger

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    try:
        if rank == 0:
            batch = _partial(["INFO", "WARN"], [3, 4], [300, 77], [3, 4])
        else:
            batch = _partial(["INFO", "ERROR"], [2, 5], [250, 400], [2, 5])
        m = DistMerger(QUERY, device="cpu")
        m.setup(batch)
        rows = m.step(batch)
        q.put((rank, rows))
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
    assert results[0] == EXPECTED
    assert results[1] == EXPECTED  # every rank sees the same final table
