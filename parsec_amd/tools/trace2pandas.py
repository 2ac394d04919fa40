#!/usr/bin/env python3
"""PABT1 binary trace -> pandas DataFrame ("ParSEC Trace Tables" analog).

Counterpart of the reference's dbp -> pandas/HDF5 pipeline
(tools/profiling/python/pbt2ptt.pyx): loads one or more rank trace files
into a tidy DataFrame (one row per event) for ad-hoc analysis, plus a
per-class summary. Optionally writes HDF5/parquet/CSV.

Usage:
    python -m parsec_amd.tools.trace2pandas trace.0 [trace.1 ...] \
        [--out tables.h5] [--summary]

As a library:
    from parsec_amd.tools.trace2pandas import load
    df = load(["trace.0", "trace.1"])   # columns: rank, tid, kind, class,
                                        # seq, t0_ns, t1_ns, dur_us, lane
"""
import argparse
import json
import struct
import sys


def load(paths):
    """Read PABT1 trace file(s) into one pandas DataFrame."""
    import pandas as pd
    rec = struct.Struct("<QQIHHQ")
    rows = []
    for path in paths:
        # rank from the trailing ".N" suffix when present
        try:
            rank = int(str(path).rsplit(".", 1)[1])
        except (IndexError, ValueError):
            rank = 0
        with open(path, "rb") as f:
            magic = f.readline().strip()
            assert magic == b"PABT1", f"not a parsec_amd trace: {magic!r}"
            header = json.loads(f.readline())
            raw = f.read()
        assert header["rec_bytes"] == rec.size
        classes = {int(k): v for k, v in header["classes"].items()}
        kinds = {int(k): v for k, v in header["kinds"].items()}
        for off in range(0, len(raw) - rec.size + 1, rec.size):
            t0, t1, tid, kind, cid, seq = rec.unpack_from(raw, off)
            rows.append((rank, tid, kinds.get(kind, str(kind)),
                         classes.get(cid, f"class{cid}"), seq, t0, t1,
                         (t1 - t0) / 1e3,
                         f"gpu-stream-{tid - 1000}" if tid >= 1000
                         else f"worker-{tid}"))
    return pd.DataFrame(rows, columns=[
        "rank", "tid", "kind", "class", "seq", "t0_ns", "t1_ns", "dur_us",
        "lane"])


def summarize(df):
    """Per-(kind, class) table: count, total/mean/max duration."""
    g = df.groupby(["kind", "class"])["dur_us"]
    out = g.agg(count="count", total_us="sum", mean_us="mean", max_us="max")
    return out.sort_values("total_us", ascending=False)


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("traces", nargs="+")
    ap.add_argument("--out", help=".h5/.parquet/.csv output path")
    ap.add_argument("--summary", action="store_true",
                    help="print the per-class summary table")
    args = ap.parse_args(argv)
    df = load(args.traces)
    print(f"{len(df)} events from {len(args.traces)} trace file(s); "
          f"kinds: {sorted(df['kind'].unique())}")
    if args.summary or not args.out:
        with __import__("pandas").option_context("display.width", 120):
            print(summarize(df).to_string())
    if args.out:
        if args.out.endswith(".h5"):
            df.to_hdf(args.out, key="events", mode="w")
        elif args.out.endswith(".parquet"):
            df.to_parquet(args.out)
        else:
            df.to_csv(args.out, index=False)
        print(f"wrote {args.out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
