#!/usr/bin/env python3
"""Run each golden query through the GPU path in a SUBPROCESS so one crash
doesn't kill the sweep; print per-case verdicts (debug utility)."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
from tests.golden_queries import GOLDEN_QUERIES  # noqa: E402

CASE_PROG = r'''
import sys, json
sys.path.insert(0, {root!r})
from parseable_amd import (GpuSession, StandardTableProvider,
                           ManifestCountResult, EmptyScanResult)
from oracle.compare import rows_equal
import json as j
g = j.load(open({root!r} + "/tests/golden/answers.json"))
entry = g["answers"][{case!r}]
fx = {case!r}.split("/")[0]
s = GpuSession()
p = StandardTableProvider({root!r} + "/tests/golden/data/" + fx, s)
plan = p.scan(entry["query"])
if isinstance(plan, (ManifestCountResult, EmptyScanResult)):
    rows = plan.rows()
else:
    rows = plan.execute_all()
ok = rows_equal(rows, entry["result"]["rows"])
print("PARITY" if ok else "MISMATCH: " + str(rows[:5]) + " vs " + str(entry["result"]["rows"][:5]))
'''

for fx, qs in GOLDEN_QUERIES.items():
    for qname, _ in qs:
        case = f"{fx}/{qname}"
        r = subprocess.run(
            [sys.executable, "-c", CASE_PROG.format(root=ROOT, case=case)],
            capture_output=True, text=True, timeout=180,
        )
        verdict = r.stdout.strip().splitlines()[-1] if r.stdout.strip() else f"rc={r.returncode}"
        if r.returncode != 0:
            verdict = f"CRASH rc={r.returncode}: " + (r.stderr.strip().splitlines()[-1] if r.stderr.strip() else "")
        print(f"{case:45s} {verdict}", flush=True)
