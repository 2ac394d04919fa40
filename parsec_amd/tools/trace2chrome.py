#!/usr/bin/env python3
"""Convert a parsec_amd binary trace (PABT1) to Chrome trace JSON.

Counterpart of the reference's PBT -> pandas -> Chrome pipeline
(tools/profiling/python/pbt2ptt.pyx + h5toctf.py) for the native trace
format written by src/profiling.cpp. Open the output in chrome://tracing
or https://ui.perfetto.dev.

Usage: python -m parsec_amd.tools.trace2chrome trace.0 [out.json]
       python -m parsec_amd.tools.trace2chrome trace.0 trace.1 ... out.json
         (multi-rank merge: rank r renders as process pid=10*r, its GPU
          device-span lanes as pid=10*r+1)
"""
import json
import struct
import sys


def convert(path, out_path=None, pid_base=0, events=None):
    own = events is None
    if events is None:
        events = []
    with open(path, "rb") as f:
        magic = f.readline().strip()
        assert magic == b"PABT1", f"not a parsec_amd trace: {magic!r}"
        header = json.loads(f.readline())
        raw = f.read()
    rec = struct.Struct("<QQIHHQ")
    assert header["rec_bytes"] == rec.size
    classes = {int(k): v for k, v in header["classes"].items()}
    kinds = {int(k): v for k, v in header["kinds"].items()}
    gpu_lanes = set()
    for off in range(0, len(raw) - rec.size + 1, rec.size):
        t0, t1, tid, kind, cid, seq = rec.unpack_from(raw, off)
        name = classes.get(cid, f"class{cid}")
        if tid >= 1000:
            gpu_lanes.add(tid)
        events.append({
            "name": f"{name}",
            "cat": kinds.get(kind, str(kind)),
            "ph": "X",
            "ts": t0 / 1e3,
            "dur": max(t1 - t0, 1) / 1e3,
            "pid": pid_base + (1 if tid >= 1000 else 0),
            "tid": tid - 1000 if tid >= 1000 else tid,
            "args": {"seq": seq},
        })
    for lane in sorted(gpu_lanes):
        events.append({"name": "thread_name", "ph": "M", "pid": pid_base + 1,
                       "tid": lane - 1000,
                       "args": {"name": f"gpu exec stream {lane - 1000}"}})
    if gpu_lanes:
        events.append({"name": "process_name", "ph": "M",
                       "pid": pid_base + 1,
                       "args": {"name": f"rank {pid_base // 10} GPU spans"}})
    if pid_base:
        events.append({"name": "process_name", "ph": "M", "pid": pid_base,
                       "args": {"name": f"rank {pid_base // 10}"}})
    if not own:
        return None, len(events)
    out = out_path or path + ".json"
    with open(out, "w") as f:
        json.dump({"traceEvents": events}, f)
    return out, len(events)


def convert_many(paths, out_path):
    """Merge several rank traces into one Chrome trace (rank r -> pid
    10r, its GPU lanes -> pid 10r+1)."""
    events = []
    for p in paths:
        try:
            rank = int(str(p).rsplit(".", 1)[1])
        except (IndexError, ValueError):
            rank = 0
        convert(p, pid_base=10 * rank, events=events)
    with open(out_path, "w") as f:
        json.dump({"traceEvents": events}, f)
    return out_path, len(events)


if __name__ == "__main__":
    if len(sys.argv) > 3:
        out, n = convert_many(sys.argv[1:-1], sys.argv[-1])
    else:
        out, n = convert(sys.argv[1],
                         sys.argv[2] if len(sys.argv) > 2 else None)
    print(f"wrote {n} events to {out}")
