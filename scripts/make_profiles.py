#!/usr/bin/env python3
"""Turn gpurun_out/prof_final rocprofv3 outputs into the committed artifacts
under profiles/:
  - r01_<wl>_kernel_stats.csv  (verbatim rocprofv3 --stats summaries)
  - r01_pmc_summary.md         (per-kernel per-launch HBM traffic)
  - pmc_traffic.json           (machine-readable; bench.py fills
                                roofline.traffic from it)

Counter units/corrections per /opt/skills/guides/MI355X_MICROARCH.md:
FETCH_SIZE / WRITE_SIZE are reported in KB; on gfx950 FETCH_SIZE reports
exactly 1/2 of the bytes of wide coalesced reads — we double it. WRITE_SIZE
is used uncorrected (calibrated against known byte counts below)."""
import argparse
import csv
import collections
import json
import os
import shutil

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(ROOT, "gpurun_out", "prof_final")
DST = os.path.join(ROOT, "profiles")


def agg_counters(path):
    agg = collections.defaultdict(float)
    cnt = collections.defaultdict(int)
    with open(path) as fh:
        for r in csv.DictReader(fh):
            k = r["Kernel_Name"].split("(")[0]
            agg[k] += float(r["Counter_Value"]) * 1024.0  # KB -> bytes
            cnt[k] += 1
    return agg, cnt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--src", default=SRC)
    ap.add_argument("--prefix", default="r01")
    ap.add_argument("--rows-map", default="",
                    help="wl=rows[,wl=rows...]: stamp the PMC scale into "
                    "pmc_traffic.json (bench quotes traffic only when its "
                    "--rows matches)")
    ap.add_argument("--merge", action="store_true",
                    help="merge into the existing pmc_traffic.json instead "
                    "of replacing it (other workloads keep their entries)")
    args = ap.parse_args()
    src = args.src
    rows_map = dict(kv.split("=") for kv in args.rows_map.split(",") if kv)

    os.makedirs(DST, exist_ok=True)
    traffic = {}
    if args.merge and os.path.exists(os.path.join(DST, "pmc_traffic.json")):
        with open(os.path.join(DST, "pmc_traffic.json")) as fh:
            traffic = json.load(fh)["traffic"]
    md = [f"# PMC summary (rocprofv3, gfx950) — {args.prefix}\n",
          "Units: FETCH_SIZE/WRITE_SIZE counters are KB; fetch doubled per the",
          "gfx950 half-reporting correction (MI355X_MICROARCH.md §HBM).",
          "Per-launch values = total / dispatch count.\n"]
    for wl in ["c1", "c2s", "c3s", "c4s"]:
        for f in ["kernel_stats", "kernel_trace"]:
            s = os.path.join(src, f"{wl}_stats_{f}.csv")
            if os.path.exists(s) and f == "kernel_stats":
                shutil.copy(s, os.path.join(DST, f"{args.prefix}_{wl}_kernel_stats.csv"))
        fpath = os.path.join(src, f"{wl}_fetch_counter_collection.csv")
        wpath = os.path.join(src, f"{wl}_write_counter_collection.csv")
        if not (os.path.exists(fpath) and os.path.exists(wpath)):
            continue
        fa, fc = agg_counters(fpath)
        wa, wc = agg_counters(wpath)
        md.append(f"\n## workload {wl}\n")
        md.append("| kernel | launches | fetch GB/launch (x2 corr) | write GB/launch |")
        md.append("|---|---|---|---|")
        traffic[wl] = {}
        if wl in rows_map:
            traffic[wl]["rows"] = int(rows_map[wl])
        for k in sorted(set(fa) | set(wa), key=lambda k: -(fa.get(k, 0))):
            if "rocclr" in k or "init_table" in k:
                continue
            n = max(fc.get(k, wc.get(k, 1)), 1)
            fpl = 2.0 * fa.get(k, 0) / n
            wpl = wa.get(k, 0) / max(wc.get(k, n), 1)
            if fpl + wpl < 1e6:
                continue
            md.append(f"| {k} | {n} | {fpl / 1e9:.3f} | {wpl / 1e9:.3f} |")
            short = ("lz4_page_decompress" if "lz4_seg" in k or "lz4_pages" in k or "lit_" in k else
                     "lz4_backrefs" if "backref" in k or "brres" in k else
                     "dict_count_fused" if "dict_count" in k else
                     "bytes_contains(LIKE)" if "contains" in k else
                     "decode+filter+groupby" if "agg" in k or "dict_pages" in k or "delta" in k or "plain" in k or "expand" in k or "def_levels" in k else k)
            traffic[wl].setdefault(short, 0.0)
            traffic[wl][short] += fpl + wpl
    with open(os.path.join(DST, f"{args.prefix}_pmc_summary.md"), "w") as fh:
        fh.write("\n".join(md) + "\n")
    with open(os.path.join(DST, "pmc_traffic.json"), "w") as fh:
        json.dump({"units": "bytes per launch (fetch x2 gfx950 correction + write)",
                   "traffic": traffic}, fh, indent=1)
    print("\n".join(md))


if __name__ == "__main__":
    main()
