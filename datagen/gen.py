"""Synthetic Parseable stream generator — writes the EXACT parquet dialect the
reference produces, plus its catalog metadata (daily manifest JSON + snapshot).

Dialect, pinned against the reference (parseablehq/parseable v2.9.5):
  - row groups of 262,144 rows           (src/cli.rs:468-474)
  - compression LZ4_RAW (codec enum 7)   (src/cli.rs:484-491, src/option.rs:83)
  - p_timestamp: TimestampMillisecond, encoding DELTA_BINARY_PACKED, rows
    sorted time-DESC, SortingColumn{descending:true, nulls_first:false}
    advertised in the footer               (src/parseable/streams.rs:705-780)
  - every other column: arrow writer defaults (dictionary/RLE_DICTIONARY with
    PLAIN fallback past the 1 MiB dict-page limit), data page v1
  - file layout <stream>/date=YYYY-MM-DD/hour=HH/minute=MM/<file>.parquet
                                           (src/utils/time.rs:216-217,249,279,349)
  - one file per minute                    (src/parseable/streams.rs:922-1001,
                                            LOCAL_SYNC_INTERVAL src/lib.rs:80)
  - daily manifest stream/date=YYYY-MM-DD/manifest.json holding File entries
    with per-column min/max TypedStatistics (src/catalog/manifest.rs:143-157,
    src/catalog/column.rs:200-206, path src/catalog/mod.rs:564-580)
  - stream.json snapshot with manifest_list of daily ManifestItem bounds
                                           (src/catalog/snapshot.rs:75-84,
                                            src/catalog/mod.rs:176-187)

Verified in-container: pyarrow 25.0 `compression='lz4'` writes codec 7
(LZ4_RAW) — the same codec parquet-rs writes for Compression::LZ4_RAW.

Configs c0..c4 follow BASELINE.json / SURVEY.md §8d.
"""

from __future__ import annotations

import json
import os
import string
from datetime import datetime, timezone

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq

ROW_GROUP_SIZE = 262_144  # src/cli.rs:468-474
BASE_TS_MS = int(datetime(2025, 9, 1, tzinfo=timezone.utc).timestamp() * 1000)
MINUTE_MS = 60_000

LEVELS = ["TRACE", "DEBUG", "INFO", "WARN", "ERROR"]  # 5 values per SURVEY §8d c1


def _zipf_probs(n: int, s: float = 1.2) -> np.ndarray:
    p = 1.0 / np.arange(1, n + 1) ** s
    return p / p.sum()


def _rand_strings(rng, n, lo, hi, alphabet=string.ascii_lowercase + string.digits):
    """n random strings with lengths uniform in [lo, hi]."""
    lens = rng.integers(lo, hi + 1, n)
    total = int(lens.sum())
    chars = np.frombuffer(
        bytes(rng.integers(0, len(alphabet), total, dtype=np.uint8)), dtype=np.uint8
    )
    lut = np.frombuffer(alphabet.encode(), dtype=np.uint8)
    flat = lut[chars].tobytes()
    out = []
    off = 0
    for L in lens:
        out.append(flat[off : off + L].decode())
        off += L
    return out


def _dict_col(rng, n, values, s=1.2):
    cum = np.cumsum(_zipf_probs(len(values), s))
    idx = np.searchsorted(cum, rng.random(n)).astype(np.int32)
    np.clip(idx, 0, len(values) - 1, out=idx)
    return pa.DictionaryArray.from_arrays(pa.array(idx), pa.array(values)).cast(
        pa.string()
    )


def _msg_col(rng, n):
    """Vectorized c3-shape message column (20-120 B, ~1% contain "error"):
    numpy byte assembly + Arrow buffers — the per-row Python string path is
    ~30x slower at the 1 B-row bench scale."""
    alphabet = (string.ascii_lowercase + string.digits).encode()
    lut = np.frombuffer(alphabet, dtype=np.uint8)
    lens = rng.integers(20, 121, n, dtype=np.int64)
    offsets = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(lens, out=offsets[1:])
    total = int(offsets[-1])
    chars = lut[rng.integers(0, len(alphabet), total, dtype=np.int64)]
    hit = rng.random(n) < 0.01
    pos_in = rng.integers(0, 15, n, dtype=np.int64)
    starts = offsets[:-1][hit] + (pos_in[hit] % np.maximum(lens[hit] - 5, 1))
    if len(starts):
        idx = (starts[:, None] + np.arange(5)).ravel()
        chars[idx] = np.tile(np.frombuffer(b"error", dtype=np.uint8),
                             len(starts))
    return pa.StringArray.from_buffers(
        n, pa.py_buffer(offsets.astype(np.int32).tobytes()),
        pa.py_buffer(chars.tobytes()))


def _minute_batch(config: str, rng: np.random.Generator, n: int, minute: int):
    """One file's rows. Timestamps descending within the minute
    (src/parseable/streams.rs:756-760: SortingColumn time DESC)."""
    t0 = BASE_TS_MS + minute * MINUTE_MS
    ts = np.sort(rng.integers(t0, t0 + MINUTE_MS, n, dtype=np.int64))[::-1].copy()
    cols = {"p_timestamp": pa.array(ts, type=pa.timestamp("ms"))}

    if config == "c0":
        # demo-stream shape per resources/ingest_demo_data.sh (CPU plumbing check)
        cols["body"] = pa.array(_rand_strings(rng, n, 60, 100))
        cols["severity_text"] = _dict_col(rng, n, [f"sev{i}" for i in range(7)])
        cols["service.name"] = _dict_col(rng, n, [f"svc-{i}" for i in range(10)])
        cols["url.path"] = _dict_col(rng, n, [f"/api/v{i}/res" for i in range(20)])
        cols["host"] = _dict_col(rng, n, [f"host-{i:04d}" for i in range(50)])
    elif config in ("c1", "c2", "c3"):
        cols["latency"] = pa.array(
            rng.integers(0, 10**6, n, dtype=np.int64), type=pa.int64()
        )
        cols["level"] = _dict_col(rng, n, LEVELS)
        cols["host"] = _dict_col(rng, n, [f"host-{i:04d}" for i in range(1000)])
        cols["f_str1"] = _dict_col(rng, n, [f"region-{i}" for i in range(16)])
        cols["f_str2"] = _dict_col(rng, n, [f"dc-{i}" for i in range(8)])
        cols["f_i64"] = pa.array(rng.integers(-(2**40), 2**40, n), type=pa.int64())
        cols["f_f64"] = pa.array(rng.random(n), type=pa.float64())
        if config == "c3":
            msgs = _rand_strings(rng, n, 20, 120)
            hit = rng.random(n) < 0.01  # 1% contain "error" 
            pos = rng.integers(0, 15, n)
            msgs = [
                (m[: pos[i] % max(1, len(m) - 5)] + "error" + m[pos[i] % max(1, len(m) - 5) :])
                if hit[i]
                else m
                for i, m in enumerate(msgs)
            ]
            cols["message"] = pa.array(msgs)
    elif config == "c3b":
        # the BASELINE c3 bench shape, slim: 1 B rows of (p_timestamp,
        # message) fit the GPU box's local disk; the LIKE scan reads only
        # the message column either way (bytes_scanned is identical)
        cols["message"] = _msg_col(rng, n)
    elif config == "c5":
        # raw-byte utf8 stress: moderate/high-cardinality string columns that
        # overflow the dictionary page (written with a small
        # dictionary_pagesize_limit) into PLAIN-fallback data pages — the
        # writer behavior of parseable/streams.rs:705-780 under dict
        # overflow, which forces the engine's row-hash group-by path.
        cols["latency"] = pa.array(
            rng.integers(0, 10**6, n, dtype=np.int64), type=pa.int64()
        )
        cols["level"] = _dict_col(rng, n, LEVELS)
        cols["host"] = _dict_col(rng, n, [f"host-{i:04d}" for i in range(1000)])
        tr = rng.integers(0, 1500, n)
        cols["trace"] = pa.array([f"tr-{v:08d}" for v in tr])
        mask = rng.random(n) < 0.3
        tagv = rng.integers(0, 800, n)
        cols["opt_tag"] = pa.array(
            [None if m else f"tag-{v:06d}" for m, v in zip(mask, tagv)])
    elif config == "c4":
        # OTel-shaped: 3 group keys + 61 sparse attribute columns, 50-90% null
        cols["service"] = _dict_col(rng, n, [f"svc-{i}" for i in range(30)])
        cols["span_kind"] = _dict_col(rng, n, ["SERVER", "CLIENT", "INTERNAL", "PRODUCER", "CONSUMER"])
        cols["status"] = _dict_col(rng, n, ["OK", "ERROR", "UNSET"])
        cols["latency"] = pa.array(rng.integers(0, 10**6, n), type=pa.int64())
        for j in range(30):
            null_frac = 0.5 + 0.4 * (j % 5) / 4
            mask = rng.random(n) < null_frac
            values = [f"attr{j}-v{k}" for k in range(12)]
            cum = np.cumsum(_zipf_probs(len(values), 1.2))
            idx = np.searchsorted(cum, rng.random(n)).astype(np.int32)
            np.clip(idx, 0, len(values) - 1, out=idx)
            ind = pa.array(idx, type=pa.int32(), mask=mask)
            cols[f"attr_s{j}"] = pa.DictionaryArray.from_arrays(
                ind, pa.array(values)).cast(pa.string())
        for j in range(31):
            null_frac = 0.5 + 0.4 * (j % 5) / 4
            mask = rng.random(n) < null_frac
            v = rng.integers(0, 10**6, n, dtype=np.int64)
            cols[f"attr_i{j}"] = pa.array(
                np.where(mask, None, v), type=pa.int64(), mask=mask
            )
    else:
        raise ValueError(config)
    return pa.table(cols)


def _ts_iso(ms: int) -> str:
    # chrono DateTime<Utc> serde format used in ManifestItem bounds
    dt = datetime.fromtimestamp(ms / 1000, tz=timezone.utc)
    return dt.strftime("%Y-%m-%dT%H:%M:%S.%f") + "Z"


def _column_stats_entry(col_meta, name):
    """Mirror of TypedStatistics JSON (externally-tagged serde enum)
    src/catalog/column.rs:26-58 + manifest Column (column.rs:200-206)."""
    st = col_meta.statistics
    stats = None
    if st is not None and st.has_min_max:
        mn, mx = st.min, st.max
        phys = col_meta.physical_type
        if phys in ("INT64", "INT32"):
            if isinstance(mn, datetime):
                mn = int(mn.timestamp() * 1000)
                mx = int(mx.timestamp() * 1000)
            stats = {"Int": {"min": int(mn), "max": int(mx)}}
        elif phys in ("FLOAT", "DOUBLE"):
            stats = {"Float": {"min": float(mn), "max": float(mx)}}
        elif phys == "BYTE_ARRAY":
            if isinstance(mn, bytes):
                mn = mn.decode("utf8", "replace")
                mx = mx.decode("utf8", "replace")
            stats = {"String": {"min": mn, "max": mx}}
        elif phys == "BOOLEAN":
            stats = {"Bool": {"min": bool(mn), "max": bool(mx)}}
    return {
        "name": name,
        "stats": stats,
        "uncompressed_size": col_meta.total_uncompressed_size,
        "compressed_size": col_meta.total_compressed_size,
    }


def _gen_one_file(args):
    (root, stream, config, rows, seed, rows_per_file, data_page_size, m,
     minute_offset, compression) = args
    return _gen_file_inner(root, stream, config, rows, seed, rows_per_file,
                           data_page_size, m, minute_offset, compression)


def gen_stream(
    root: str,
    stream: str = "bench",
    config: str = "c1",
    rows: int = 1_000_000,
    seed: int = 42,
    rows_per_file: int = ROW_GROUP_SIZE,
    data_page_size: int | None = None,
    quiet: bool = True,
    workers: int = 1,
    minute_offset: int = 0,
    compression: str = "lz4",
    manifest_codec: str | None = None,
):
    """Write a synthetic stream. Returns dict with file list + manifest paths.
    workers > 1 parallelizes per-file generation (deterministic: per-file rng
    streams); minute_offset shifts the time range (per-rank shards)."""
    n_files = (rows + rows_per_file - 1) // rows_per_file
    stream_dir = os.path.join(root, stream)
    os.makedirs(stream_dir, exist_ok=True)
    manifest_files = []  # File entries (src/catalog/manifest.rs:143-152)
    file_paths = []

    if workers > 1 and n_files > 2:
        from concurrent.futures import ProcessPoolExecutor

        argl = [
            (root, stream, config, rows, seed, rows_per_file, data_page_size,
             m, minute_offset, compression)
            for m in range(n_files)
        ]
        with ProcessPoolExecutor(max_workers=workers) as ex:
            for i, (abs_path, rel_path, entry) in enumerate(ex.map(_gen_one_file, argl, chunksize=4)):
                manifest_files.append(entry)
                file_paths.append(abs_path)
                if not quiet and (i % 50 == 0):
                    print(f"  wrote {i + 1}/{n_files} files", flush=True)
        return _finish_stream(root, stream, stream_dir, manifest_files,
                              file_paths, rows, config, manifest_codec)

    for m in range(n_files):
        abs_path, rel_path, entry = _gen_file_inner(
            root, stream, config, rows, seed, rows_per_file, data_page_size,
            m, minute_offset, compression)
        manifest_files.append(entry)
        file_paths.append(abs_path)
        if not quiet and (m % 50 == 0):
            print(f"  wrote {m + 1}/{n_files} files", flush=True)
    return _finish_stream(root, stream, stream_dir, manifest_files, file_paths,
                          rows, config, manifest_codec)


def _gen_file_inner(root, stream, config, rows, seed, rows_per_file,
                    data_page_size, m, minute_offset, compression="lz4"):
    if True:
        n = min(rows_per_file, rows - m * rows_per_file)
        rng = np.random.default_rng([seed, m])  # per-file stream: parallel-safe
        tbl = _minute_batch(config, rng, n, m + minute_offset)
        t0 = BASE_TS_MS + (m + minute_offset) * MINUTE_MS
        dt = datetime.fromtimestamp(t0 / 1000, tz=timezone.utc)
        rel_dir = (
            f"{stream}/date={dt:%Y-%m-%d}/hour={dt:%H}/minute={dt:%M}"
        )
        os.makedirs(os.path.join(root, rel_dir), exist_ok=True)
        fname = f"data.{m:06d}.parquet"
        rel_path = f"{rel_dir}/{fname}"
        abs_path = os.path.join(root, rel_path)
        kw = {}
        if data_page_size:
            kw["data_page_size"] = data_page_size
        if config == "c5":
            # force dict-page overflow -> PLAIN fallback mid-chunk
            kw["dictionary_pagesize_limit"] = 4096
        ts_idx = tbl.schema.get_field_index("p_timestamp")
        pq.write_table(
            tbl,
            abs_path,
            row_group_size=ROW_GROUP_SIZE,
            compression=compression,      # "lz4" = codec 7 LZ4_RAW (default,
                                          # cli.rs:484-491); "snappy" = the
                                          # test-compose codec
            use_dictionary=[c for c in tbl.column_names if c != "p_timestamp"],
            column_encoding={"p_timestamp": "DELTA_BINARY_PACKED"},
            data_page_version="1.0",
            write_statistics=True,
            sorting_columns=[pq.SortingColumn(ts_idx, descending=True, nulls_first=False)],
            **kw,
        )
        md = pq.read_metadata(abs_path)
        cols = {}
        ing = 0
        for rg_i in range(md.num_row_groups):
            rg = md.row_group(rg_i)
            ing += rg.total_byte_size
            for ci in range(rg.num_columns):
                cm = rg.column(ci)
                name = cm.path_in_schema
                e = _column_stats_entry(cm, name)
                if name in cols:
                    prev = cols[name]
                    prev["uncompressed_size"] += e["uncompressed_size"]
                    prev["compressed_size"] += e["compressed_size"]
                    if prev["stats"] and e["stats"]:
                        (k1, v1), (k2, v2) = (
                            next(iter(prev["stats"].items())),
                            next(iter(e["stats"].items())),
                        )
                        if k1 == k2:
                            v1["min"] = min(v1["min"], v2["min"])
                            v1["max"] = max(v1["max"], v2["max"])
                    else:
                        prev["stats"] = None
                else:
                    cols[name] = e
        entry = {
            "file_path": rel_path,
            "num_rows": md.num_rows,
            "file_size": os.path.getsize(abs_path),
            "ingestion_size": ing,
            "columns": list(cols.values()),
            "sort_order_id": [
                {
                    "field_name": "p_timestamp",
                    "sort_kind": "AtTimestamp",
                    "descending": True,
                }
            ],
        }
        return abs_path, rel_path, entry


def _zstd_compress(data: bytes, level: int = 3) -> bytes:
    # same library + level as the reference's manifest codec
    # (catalog/manifest.rs ZSTD_LEVEL = 3, via the zstd crate -> libzstd)
    import ctypes

    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_compressBound.restype = ctypes.c_size_t
    z.ZSTD_compress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint
    bound = z.ZSTD_compressBound(len(data))
    buf = ctypes.create_string_buffer(bound)
    n = z.ZSTD_compress(buf, bound, data, len(data), level)
    if z.ZSTD_isError(n):
        raise RuntimeError("zstd compress failed")
    return buf.raw[:n]


def _finish_stream(root, stream, stream_dir, manifest_files, file_paths, rows,
                   config, manifest_codec=None):
    # daily manifests (partition bounds = whole UTC day, src/catalog/mod.rs:176-187)
    by_day = {}
    for f in manifest_files:
        day = f["file_path"].split("date=")[1].split("/")[0]
        by_day.setdefault(day, []).append(f)
    manifest_list = []
    for day, files in sorted(by_day.items()):
        rel_mpath = f"{stream}/date={day}/manifest.json"
        mpath = os.path.join(root, rel_mpath)
        mbytes = json.dumps({"version": "v2", "files": files}).encode()
        if manifest_codec == "zstd":
            mbytes = _zstd_compress(mbytes)
        with open(mpath, "wb") as fh:
            fh.write(mbytes)
        d0 = datetime.strptime(day, "%Y-%m-%d").replace(tzinfo=timezone.utc)
        lo = int(d0.timestamp() * 1000)
        manifest_list.append(
            {
                "manifest_path": rel_mpath,
                "time_lower_bound": _ts_iso(lo),
                "time_upper_bound": _ts_iso(lo + 86_400_000 - 1),
                "events_ingested": sum(f["num_rows"] for f in files),
                "ingestion_size": sum(f["ingestion_size"] for f in files),
                "storage_size": sum(f["file_size"] for f in files),
            }
        )
    snap_path = os.path.join(stream_dir, "stream.json")
    with open(snap_path, "w") as fh:
        json.dump(
            {
                "version": "v6",
                "objectstore-format": "v6",
                "stream_type": "UserDefined",
                "snapshot": {"version": "v2", "manifest_list": manifest_list},
            },
            fh,
        )
    return {
        "stream_dir": stream_dir,
        "snapshot": snap_path,
        "files": file_paths,
        "manifest_files": manifest_files,
        "rows": rows,
        "config": config,
    }


def gen_staging(
    staging_dir: str,
    config: str = "c1",
    rows: int = 30_000,
    seed: int = 77,
    n_arrows: int = 2,
    n_parquet: int = 1,
    start_minute: int = 10_000,
):
    """Staging-window data, as the reference's staging directory holds it
    between ingest and upload (src/parseable/staging/): per-minute `.arrows`
    Arrow-IPC stream files plus `.parquet` files already converted
    (parseable/streams.rs:922-1001) but not yet uploaded/manifested.
    Timestamps start at BASE_TS_MS + start_minute minutes; callers set the
    provider's now_ms inside [min_ts, max_ts + window)."""
    os.makedirs(staging_dir, exist_ok=True)
    total = n_arrows + n_parquet
    per = max(1, rows // max(total, 1))
    out = {"arrows": [], "parquet": []}
    for i in range(total):
        rng = np.random.default_rng([seed, 9000 + i])
        tbl = _minute_batch(config, rng, per, start_minute + i)
        if i < n_parquet:
            p = os.path.join(staging_dir, f"staged.{i:04d}.parquet")
            ts_idx = tbl.schema.get_field_index("p_timestamp")
            pq.write_table(
                tbl, p, row_group_size=ROW_GROUP_SIZE, compression="lz4",
                use_dictionary=[c for c in tbl.column_names if c != "p_timestamp"],
                column_encoding={"p_timestamp": "DELTA_BINARY_PACKED"},
                data_page_version="1.0", write_statistics=True,
                sorting_columns=[pq.SortingColumn(ts_idx, descending=True,
                                                  nulls_first=False)],
            )
            out["parquet"].append(p)
        else:
            p = os.path.join(staging_dir, f"{start_minute + i}.data.arrows")
            with pa.ipc.new_stream(p, tbl.schema) as w:
                for b in tbl.to_batches(max_chunksize=8192):
                    w.write_batch(b)
            out["arrows"].append(p)
    out["min_ts"] = BASE_TS_MS + start_minute * MINUTE_MS
    out["max_ts"] = BASE_TS_MS + (start_minute + total) * MINUTE_MS
    return out


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--root", required=True)
    ap.add_argument("--stream", default="bench")
    ap.add_argument("--config", default="c1")
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--rows-per-file", type=int, default=ROW_GROUP_SIZE)
    a = ap.parse_args()
    info = gen_stream(
        a.root, a.stream, a.config, a.rows, a.seed, a.rows_per_file, quiet=False
    )
    print(json.dumps({k: info[k] for k in ("stream_dir", "snapshot", "rows")}, indent=2))
