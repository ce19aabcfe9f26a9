#!/usr/bin/env python3
"""Generate the committed golden fixtures: parquet inputs (Parseable dialect)
under tests/golden/data/<fixture>/ and expected answers in
tests/golden/answers.json, computed by the pyarrow/numpy oracle and
cross-checked against pyarrow Acero at generation time.

Run from the repo root:  python3 scripts/gen_golden.py
Committed outputs are the parity anchor for the C oracle and the GPU path
(/root/reference is absent on the GPU box — nothing at run time may read it).
"""
import json
import os
import shutil
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from datagen.gen import gen_stream  # noqa: E402
from oracle import query_oracle as qo  # noqa: E402
from oracle.compare import FLOAT_RTOL, rows_equal  # noqa: E402
from tests.golden_queries import GOLDEN_FIXTURES, GOLDEN_QUERIES  # noqa: E402

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GDIR = os.path.join(ROOT, "tests", "golden", "data")


def main():
    if os.path.isdir(GDIR):
        shutil.rmtree(GDIR)
    os.makedirs(GDIR)
    answers = {}
    manifest = {}
    for name, params in GOLDEN_FIXTURES.items():
        info = gen_stream(GDIR, name, **params)
        rel_files = [os.path.relpath(f, GDIR) for f in info["files"]]
        manifest[name] = {"files": rel_files, **{k: v for k, v in params.items()}}
        for qname, query in GOLDEN_QUERIES.get(name, []):
            r1 = qo.execute(info["files"], query)
            r2 = qo.execute_acero(info["files"], query)
            assert rows_equal(r1["rows"], r2["rows"], float_rtol=FLOAT_RTOL), (
                f"oracle disagreement on {name}/{qname}:\n{r1['rows']}\nvs\n{r2['rows']}"
            )
            answers[f"{name}/{qname}"] = {"query": query, "result": r1}
            print(f"{name}/{qname}: {len(r1['rows'])} groups OK")
    with open(os.path.join(ROOT, "tests", "golden", "answers.json"), "w") as fh:
        json.dump({"fixtures": manifest, "answers": answers}, fh, indent=1)
    total = sum(
        os.path.getsize(os.path.join(dp, f))
        for dp, _, fs in os.walk(GDIR)
        for f in fs
    )
    print(f"golden data: {total / 1e6:.1f} MB")


if __name__ == "__main__":
    main()
