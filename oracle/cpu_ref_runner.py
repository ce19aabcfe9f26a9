"""Run the scalar C oracle CLI (oracle/_ref/cpu_ref) on a query-IR dict and
parse its output into the same normalized rows the Python oracle returns.
ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py)."""

from __future__ import annotations

import os
import subprocess

ORACLE_DIR = os.path.dirname(os.path.abspath(__file__))
CPU_REF = os.path.join(ORACLE_DIR, "_ref", "cpu_ref")


def build():
    subprocess.run(["make", "-C", ORACLE_DIR], check=True, capture_output=True)


def available() -> bool:
    return os.path.exists(CPU_REF)


def _query_args(query: dict) -> list[str]:
    args = []
    tr = query.get("time_range")
    if tr is not None:
        args += ["--time", f"{tr[0]},{tr[1]}"]
    gb = query.get("group_by", [])
    if gb:
        args += ["--group-by", ",".join(gb)]
    for a in query["select"]:
        if a["agg"] == "count_star":
            args += ["--agg", "count_star"]
        else:
            args += ["--agg", f"{a['agg']}:{a['col']}"]
    for p in query.get("preds", []):
        if p["op"] == "between":
            args += ["--pred", f"between:{p['col']}:i64:{p['lo']},{p['hi']}"]
        elif p["op"] == "contains":
            args += ["--pred", f"contains:{p['col']}:str:{p['lit']}"]
        else:
            lit = p["lit"]
            kind = "str" if isinstance(lit, str) else ("f64" if isinstance(lit, float) else "i64")
            args += ["--pred", f"{p['op']}:{p['col']}:{kind}:{lit}"]
    return args


def execute(files: list[str], query: dict, timeout=300) -> dict:
    if not available():
        build()
    cmd = [CPU_REF] + _query_args(query) + list(files)
    out = subprocess.run(cmd, check=True, capture_output=True, timeout=timeout)
    n_keys = len(query.get("group_by", []))
    aggs = query["select"]
    rows = []
    for line in out.stdout.decode().splitlines():
        parts = line.split("\t")
        row = []
        for i, v in enumerate(parts):
            if v == "\\N":
                row.append(None)
            elif i < n_keys:
                row.append(v)
            else:
                row.append(float(v) if ("." in v or "e" in v or "inf" in v or "nan" in v) else int(v))
        rows.append(row)
    return {"rows": rows}
