#!/usr/bin/env python3
"""Randomized query-parity fuzz: generate seeded random queries over small
c1/c3/c4 streams and compare the GPU path against the oracle, case by case.
Usage: python scripts/gpu_fuzz.py [--n 60] [--seed 1] [--cpu-check]
--cpu-check validates the generated query space oracle-vs-acero only (no
GPU; run HERE before spending box time)."""
import argparse
import os
import random
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from datagen.gen import BASE_TS_MS, MINUTE_MS, gen_stream  # noqa: E402
from oracle import query_oracle as qo  # noqa: E402
from oracle.compare import FLOAT_RTOL, rows_equal  # noqa: E402

STREAMS = {
    "c1": {"rows": 500_000, "keys": ["level", "host", "f_str1", "f_str2"],
           "i64": ["latency", "f_i64"], "f64": ["f_f64"], "contains": None,
           "ikeys": ["latency", "f_i64", "f_f64"]},
    "c3": {"rows": 300_000, "keys": ["level", "host", "f_str1"],
           "i64": ["latency", "f_i64"], "f64": ["f_f64"],
           "contains": "message"},
    "c4": {"rows": 300_000,
           "keys": ["service", "span_kind", "status", "attr_s1", "attr_s7"],
           "i64": ["latency", "attr_i0", "attr_i3", "attr_i11"], "f64": [],
           "contains": None, "ikeys": ["attr_i5", "attr_i11"]},
    # dict-overflow PLAIN-fallback utf8 (raw-byte hash group-by path)
    "c5": {"rows": 300_000, "keys": ["level", "host", "trace", "opt_tag"],
           "i64": ["latency"], "f64": [], "contains": "trace"},
    # the c1 shape written with SNAPPY pages + zstd manifests (the
    # reference's test-compose codec / compressed-manifest combination)
    "c6": {"rows": 300_000, "keys": ["level", "host", "f_str1", "f_str2"],
           "i64": ["latency", "f_i64"], "f64": ["f_f64"], "contains": None,
           "gen": {"config": "c1", "compression": "snappy",
                   "manifest_codec": "zstd"}},
}
KEY_VALUES = {
    "level": ["TRACE", "DEBUG", "INFO", "WARN", "ERROR"],
    "host": [f"host-{i:04d}" for i in range(1000)],
    "f_str1": [f"region-{i}" for i in range(16)],
    "f_str2": [f"dc-{i}" for i in range(8)],
    "service": [f"svc-{i}" for i in range(30)],
    "span_kind": ["SERVER", "CLIENT", "INTERNAL", "PRODUCER", "CONSUMER"],
    "status": ["OK", "ERROR", "UNSET"],
    "attr_s1": [f"attr1-v{k}" for k in range(12)],
    "attr_s7": [f"attr7-v{k}" for k in range(12)],
    "trace": [f"tr-{v:08d}" for v in range(1500)],
    "opt_tag": [f"tag-{v:06d}" for v in range(800)],
}


def gen_query(rng: random.Random, cfg: dict, n_rows: int):
    q = {}
    keys = rng.sample(cfg["keys"], rng.randint(0, min(3, len(cfg["keys"]))))
    if keys and rng.random() < 0.15:
        keys[0] = {"bin": "p_timestamp",
                   "stride_ms": rng.choice([60_000, 300_000])}
    if rng.random() < 0.2 and cfg.get("ikeys") and len(keys) < 3:
        keys.append(rng.choice(cfg["ikeys"]))  # numeric group key
    q["group_by"] = keys
    aggs = [{"agg": "count_star"}]
    numcols = cfg["i64"] + cfg["f64"] + ["p_timestamp"]
    for _ in range(rng.randint(0, 3)):
        col = rng.choice(numcols)
        op = rng.choice(["sum", "avg", "min", "max", "count"])
        aggs.append({"agg": op, "col": col})
    if rng.random() < 0.25 and cfg["keys"]:
        aggs.append({"agg": rng.choice(["min", "max"]),
                     "col": rng.choice(cfg["keys"])})
    if rng.random() < 0.3:
        aggs.append({"agg": "count", "col": rng.choice(cfg["keys"])})
    seen = set()
    dedup = []
    for a in aggs:  # duplicate (op,col) pairs break the acero check leg
        sig = (a["agg"], a.get("col"))
        if sig not in seen:
            seen.add(sig)
            dedup.append(a)
    q["select"] = dedup
    preds = []
    for _ in range(rng.randint(0, 2)):
        kind = rng.random()
        if kind < 0.45:
            col = rng.choice(cfg["i64"]) if cfg["i64"] else None
            if col:
                lit = rng.randint(0, 10**6)
                preds.append({"col": col,
                              "op": rng.choice(["lt", "le", "gt", "ge"]),
                              "lit": lit})
        elif kind < 0.7 and cfg["f64"]:
            preds.append({"col": rng.choice(cfg["f64"]),
                          "op": rng.choice(["lt", "gt"]),
                          "lit": rng.random()})
        elif kind < 0.9:
            col = rng.choice(cfg["keys"])
            preds.append({"col": col, "op": rng.choice(["eq", "ne"]),
                          "lit": rng.choice(KEY_VALUES[col])})
        elif cfg["contains"]:
            preds.append({"col": cfg["contains"], "op": "contains",
                          "lit": rng.choice(["error", "qx", "ab", "zzz"])})
    if cfg["contains"] and rng.random() < 0.15:
        for lit in rng.sample(["error", "qx", "ab", "ror", "err"],
                              rng.randint(1, 2)):
            preds.append({"col": cfg["contains"], "op": "contains",
                          "lit": lit})
    if rng.random() < 0.2 and cfg["i64"]:
        col = rng.choice(cfg["i64"])
        lo = rng.randint(0, 800_000)
        preds.append({"col": col, "op": "between", "lo": lo,
                      "hi": lo + rng.randint(1, 400_000)})
    q["preds"] = preds
    if rng.random() < 0.35:
        n_files = (n_rows + 262_143) // 262_144
        span = n_files * MINUTE_MS
        lo = BASE_TS_MS + rng.randint(0, span // 2)
        hi = lo + rng.randint(MINUTE_MS // 4, span)
        q["time_range"] = (lo, hi)
    if rng.random() < 0.12:  # top-k projection scan instead of aggregation
        q.pop("select", None)
        q["group_by"] = []
        cols = ["p_timestamp"] + rng.sample(cfg["keys"] + cfg["i64"],
                                            rng.randint(1, 3))
        q["select_cols"] = cols
        q["limit"] = rng.choice([5, 50, 500])
        q["order_by"] = {"col": "p_timestamp", "desc": True}
    return q


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=60)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--cpu-check", action="store_true")
    ap.add_argument("--data-dir", default="/tmp/gpuq_fuzz")
    ap.add_argument("--only", type=int, default=-1,
                    help="run only case #N (rng stream still advances)")
    ap.add_argument("--isolate", action="store_true",
                    help="one subprocess per case: a GPU fault kills only "
                         "that case and is reported with its query")
    args = ap.parse_args()

    if args.isolate:
        import subprocess

        bad = 0
        for t in range(args.n):
            r = subprocess.run(
                [sys.executable, "-u", os.path.abspath(__file__),
                 "--n", str(args.n), "--seed", str(args.seed),
                 "--only", str(t), "--data-dir", args.data_dir],
                capture_output=True, text=True, timeout=600)
            out = (r.stdout + r.stderr).strip().splitlines()
            line = next((l for l in out if l.startswith(("ok", "MISMATCH"))),
                        None)
            if r.returncode != 0 or line is None or "MISMATCH" in (line or ""):
                bad += 1
                print(f"FAIL #{t} rc={r.returncode}")
                for l in out[-12:]:
                    print("   ", l)
            else:
                print(line, flush=True)
        print(f"fuzz done (isolated): {args.n - bad}/{args.n} clean")
        return 1 if bad else 0

    streams = {}
    for cfg_name, cfg in STREAMS.items():
        root = os.path.join(args.data_dir, cfg_name)
        info_path = os.path.join(root, "stream", "stream.json")
        if not os.path.exists(info_path):
            g = cfg.get("gen", {})
            gen_stream(root, "stream", g.get("config", cfg_name),
                       rows=cfg["rows"], seed=5, workers=4,
                       **{k: v for k, v in g.items() if k != "config"})
        import glob

        files = sorted(glob.glob(os.path.join(root, "stream", "**",
                                              "*.parquet"), recursive=True))
        streams[cfg_name] = (os.path.join(root, "stream"), files, cfg)

    sess = None
    if not args.cpu_check:
        from parseable_amd import GpuSession

        sess = GpuSession(device_mask=1)
    from parseable_amd.provider import StandardTableProvider, Query

    rng = random.Random(args.seed)
    bad = 0
    for t in range(args.n):
        cfg_name = rng.choice(list(STREAMS.keys()))
        stream_dir, files, cfg = streams[cfg_name]
        q = gen_query(rng, cfg, cfg["rows"])
        if args.only >= 0 and t != args.only:
            continue
        print(f"case #{t} [{cfg_name}] {q}", flush=True)
        want = qo.execute(files, dict(q))["rows"]
        if args.cpu_check:
            if any(isinstance(g, dict) for g in q["group_by"]) or \
                    q.get("select_cols"):
                print(f"skip #{t} [{cfg_name}] (acero leg lacks "
                      f"DATE_BIN/projection)")
                continue
            got = qo.execute_acero(files, dict(q))["rows"]
            label = "acero"
        else:
            got, _ = Query(StandardTableProvider(stream_dir, sess)).execute(dict(q))
            label = "gpu"
        if q.get("select_cols"):
            ts_i = q["select_cols"].index("p_timestamp")
            ok = [r[ts_i] for r in got] == [r[ts_i] for r in want]
        else:
            # GPU vs oracle: the 1-ULP gate (both sides correctly rounded).
            # Acero cross-checks: rtol — its float sums/means are
            # order-dependent (same rule as every other Acero comparison).
            ok = rows_equal(got, want,
                            float_rtol=FLOAT_RTOL if args.cpu_check else None)
        if not ok:
            bad += 1
            print(f"MISMATCH #{t} [{cfg_name}] {q}")
            print(f"  {label}: {got[:4]}")
            print(f"  oracle: {want[:4]}")
        else:
            print(f"ok #{t} [{cfg_name}] keys={len(q['group_by'])} "
                  f"aggs={len(q.get('select', []))} "
                  f"preds={len(q['preds'])} rows={len(want)}")
    print(f"fuzz done: {args.n - bad}/{args.n} matched")
    return 1 if bad else 0


if __name__ == "__main__":
    sys.exit(main())
