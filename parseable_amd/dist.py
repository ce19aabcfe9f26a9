"""Cross-GPU Final merge of partial aggregate tables over torch.distributed
(RCCL on ROCm — backend "nccl" — over xGMI; gloo on CPU for tests).

Design per SURVEY.md §8e: row-group shards are embarrassingly parallel; the
ONLY data-path exchange is this merge of fixed-width aggregate tables —
KB..MB payloads, latency-bound on xGMI. Key spaces can differ per rank (each
rank's plan builds its global dictionary from its own shard's dict pages), so
a one-time all_gather_object agrees on the union key space (setup, untimed);
every step then reduces dense tensors: SUM for presence/counts/sums, MIN/MAX
for min/max — mirroring the reference's AggregateExec Partial->Final split
(SURVEY.md §3a step 7)."""

from __future__ import annotations

import torch
import torch.distributed as dist

I64_MAX = 2**63 - 1
I64_MIN = -(2**63)


def _batch_rows(batch, n_keys, n_aggs):
    """partial batch (C-ABI schema: keys..., __presence, agg{i}, agg{i}_count)
    -> {key_tuple: (presence, [(val, cnt), ...])}"""
    out = {}
    if batch is None or batch.num_rows == 0:
        return out
    from .provider import _col_list

    cols = [_col_list(batch.column(i)) for i in range(batch.num_columns)]
    for r in range(batch.num_rows):
        key = tuple(cols[k][r] for k in range(n_keys))
        presence = cols[n_keys][r]
        aggs = []
        for i in range(n_aggs):
            aggs.append((cols[n_keys + 1 + 2 * i][r], cols[n_keys + 2 + 2 * i][r]))
        out[key] = (presence, aggs)
    return out


class DistMerger:
    """Final-merge helper. setup() once (untimed; agrees the key space),
    then step(batch) per timed iteration (dense all_reduce merge)."""

    def __init__(self, query: dict, device: str = "cpu"):
        self.query = query
        self.group_by = query.get("group_by", [])
        self.aggs = query["select"]
        self.device = device
        self.keyspace: list[tuple] | None = None
        self.key_index: dict[tuple, int] = {}
        self.is_f64 = [False] * len(self.aggs)
        self.is_str = [False] * len(self.aggs)

    @property
    def world(self):
        return dist.get_world_size() if dist.is_initialized() else 1

    def setup(self, batch):
        import pyarrow as pa

        if batch is not None:
            nk = len(self.group_by)
            for i in range(len(self.aggs)):
                t = batch.schema.field(nk + 1 + 2 * i).type
                self.is_f64[i] = pa.types.is_floating(t)
                self.is_str[i] = pa.types.is_string(t)
        local = sorted(_batch_rows(batch, len(self.group_by), len(self.aggs)).keys(),
                       key=lambda k: tuple(((1, "") if v is None else (0, v)) for v in k))
        if self.world > 1:
            # ship the per-agg types with the key space: a rank whose shard
            # produced no rows (batch None) must still agree on f64/str slots,
            # or it would emit int-slot values and skip the string-merge path
            # while data-bearing ranks take it.
            gathered: list = [None] * self.world
            dist.all_gather_object(gathered, (local, self.is_f64, self.is_str))
            allk = set()
            for keys, f64s, strs in gathered:
                allk.update(keys)
                self.is_f64 = [a or b for a, b in zip(self.is_f64, f64s)]
                self.is_str = [a or b for a, b in zip(self.is_str, strs)]
        else:
            allk = set(local)
        self.keyspace = sorted(allk, key=lambda k: tuple(((1, "") if v is None else (0, v)) for v in k))
        self.key_index = {k: i for i, k in enumerate(self.keyspace)}

    def step(self, batch) -> list:
        """Merge this rank's partial with every other rank's; returns the
        final rows (same normalized form as the oracle). Vectorized with
        numpy — per-row Python/torch ops cost ~20ms/step at 1000 groups."""
        import numpy as np

        assert self.keyspace is not None, "call setup() first"
        if self.world == 1:
            # single rank: the partial IS the final table — skip the dense
            # keyspace/tensor machinery entirely
            from .provider import merge_partials

            return merge_partials([batch], self.query)
        G = max(len(self.keyspace), 1)
        n_aggs = len(self.aggs)
        # utf8 min/max: rare, tiny payload (one string per live group) —
        # merged via an object gather alongside the dense tensor reduces.
        str_minmax: list[dict | None] = [None] * n_aggs
        for i, a in enumerate(self.aggs):
            if a["agg"] in ("min", "max") and self.is_str[i]:
                str_minmax[i] = {}

        np_presence = np.zeros(G, dtype=np.int64)
        np_counts = np.zeros((G, n_aggs), dtype=np.int64)
        np_sums_i = np.zeros((G, n_aggs), dtype=np.int64)
        np_sums_f = np.zeros((G, n_aggs), dtype=np.float64)
        np_mins = np.full((G, n_aggs), I64_MAX, dtype=np.int64)
        np_maxs = np.full((G, n_aggs), I64_MIN, dtype=np.int64)
        np_mins_f = np.full((G, n_aggs), np.inf)
        np_maxs_f = np.full((G, n_aggs), -np.inf)

        if batch is not None and batch.num_rows:
            nk = len(self.group_by)
            from .provider import _col_list

            keycols = [_col_list(batch.column(k)) for k in range(nk)]
            keys = list(zip(*keycols)) if nk else [()] * batch.num_rows
            try:
                gis = np.fromiter((self.key_index[k] for k in keys),
                                  dtype=np.int64, count=len(keys))
            except KeyError as e:
                raise RuntimeError(f"key {e} not in agreed key space")
            np.add.at(np_presence, gis,
                      batch.column(nk).to_numpy(zero_copy_only=False).astype(np.int64))
            for i, a in enumerate(self.aggs):
                vcol = batch.column(nk + 1 + 2 * i)
                ccol = batch.column(nk + 2 + 2 * i).to_numpy(zero_copy_only=False).astype(np.int64)
                if a["agg"] in ("count_star", "count"):
                    np.add.at(np_counts[:, i], gis,
                              vcol.to_numpy(zero_copy_only=False).astype(np.int64))
                    continue
                np.add.at(np_counts[:, i], gis, ccol)
                has = ccol > 0
                if not has.any():
                    continue
                g2 = gis[has]
                if str_minmax[i] is not None:
                    d = str_minmax[i]
                    vals = vcol.to_pylist()
                    for gi_, v_ in zip(g2.tolist(),
                                       (vals[j] for j in np.nonzero(has)[0])):
                        if v_ is None:
                            continue
                        cur = d.get(gi_)
                        if cur is None:
                            d[gi_] = v_
                        elif a["agg"] == "min":
                            d[gi_] = min(cur, v_)
                        else:
                            d[gi_] = max(cur, v_)
                    continue
                if self.is_f64[i]:
                    v = vcol.to_numpy(zero_copy_only=False).astype(np.float64)[has]
                    if a["agg"] in ("sum", "avg"):
                        np.add.at(np_sums_f[:, i], g2, v)
                    elif a["agg"] == "min":
                        np.minimum.at(np_mins_f[:, i], g2, v)
                    else:
                        np.maximum.at(np_maxs_f[:, i], g2, v)
                else:
                    import pyarrow as pa

                    fill = {"sum": 0, "avg": 0, "min": I64_MAX,
                            "max": I64_MIN}[a["agg"]]
                    v = vcol.fill_null(fill).to_numpy(zero_copy_only=False).astype(np.int64)[has]
                    if a["agg"] in ("sum", "avg"):
                        np.add.at(np_sums_i[:, i], g2, v)
                    elif a["agg"] == "min":
                        np.minimum.at(np_mins[:, i], g2, v)
                    else:
                        np.maximum.at(np_maxs[:, i], g2, v)

        dev = self.device
        presence = torch.from_numpy(np_presence).to(dev)
        counts = torch.from_numpy(np_counts).to(dev)
        sums_i = torch.from_numpy(np_sums_i).to(dev)
        sums_f = torch.from_numpy(np_sums_f).to(dev)
        mins = torch.from_numpy(np_mins).to(dev)
        maxs = torch.from_numpy(np_maxs).to(dev)
        mins_f = torch.from_numpy(np_mins_f).to(dev)
        maxs_f = torch.from_numpy(np_maxs_f).to(dev)

        if self.world > 1:
            dist.all_reduce(presence, op=dist.ReduceOp.SUM)
            dist.all_reduce(counts, op=dist.ReduceOp.SUM)
            if any(a["agg"] in ("sum", "avg") for a in self.aggs):
                dist.all_reduce(sums_i, op=dist.ReduceOp.SUM)
                dist.all_reduce(sums_f, op=dist.ReduceOp.SUM)
            if any(a["agg"] == "min" for a in self.aggs):
                dist.all_reduce(mins, op=dist.ReduceOp.MIN)
                dist.all_reduce(mins_f, op=dist.ReduceOp.MIN)
            if any(a["agg"] == "max" for a in self.aggs):
                dist.all_reduce(maxs, op=dist.ReduceOp.MAX)
                dist.all_reduce(maxs_f, op=dist.ReduceOp.MAX)
            if any(d is not None for d in str_minmax):
                gathered: list = [None] * self.world
                dist.all_gather_object(gathered, str_minmax)
                for i, a in enumerate(self.aggs):
                    if str_minmax[i] is None:
                        continue
                    merged = {}
                    pick = min if a["agg"] == "min" else max
                    for per_rank in gathered:
                        for gi, v in per_rank[i].items():
                            cur = merged.get(gi)
                            merged[gi] = v if cur is None else pick(cur, v)
                    str_minmax[i] = merged

        presence = presence.cpu()
        counts = counts.cpu()
        sums_i, sums_f = sums_i.cpu(), sums_f.cpu()
        mins, maxs = mins.cpu(), maxs.cpu()
        mins_f, maxs_f = mins_f.cpu(), maxs_f.cpu()
        out = []
        for gi, key in enumerate(self.keyspace):
            if self.group_by and presence[gi].item() == 0:
                continue
            row = list(key)
            for i, a in enumerate(self.aggs):
                c = counts[gi, i].item()
                if a["agg"] in ("count_star", "count"):
                    row.append(c)
                elif c == 0:
                    row.append(None)
                elif a["agg"] == "sum":
                    row.append(sums_f[gi, i].item() if self.is_f64[i]
                               else sums_i[gi, i].item())
                elif a["agg"] == "avg":
                    sv = (sums_f[gi, i].item() if self.is_f64[i]
                          else sums_i[gi, i].item())
                    row.append(sv / c)
                elif str_minmax[i] is not None:
                    row.append(str_minmax[i].get(gi))
                elif a["agg"] == "min":
                    row.append(mins_f[gi, i].item() if self.is_f64[i]
                               else mins[gi, i].item())
                elif a["agg"] == "max":
                    row.append(maxs_f[gi, i].item() if self.is_f64[i]
                               else maxs[gi, i].item())
            out.append(row)
        if not self.group_by and not out:
            out = [[0 if a["agg"] in ("count_star", "count") else None for a in self.aggs]]
        return out
