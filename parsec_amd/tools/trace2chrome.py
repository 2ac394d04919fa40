#!/usr/bin/env python3
"""Convert a parsec_amd binary trace (PABT1) to Chrome trace JSON.

Counterpart of the reference's PBT -> pandas -> Chrome pipeline
(tools/profiling/python/pbt2ptt.pyx + h5toctf.py) for the native trace
format written by src/profiling.cpp. Open the output in chrome://tracing
or https://ui.perfetto.dev.

Usage: python -m parsec_amd.tools.trace2chrome trace.0 [out.json]
"""
import json
import struct
import sys


def convert(path, out_path=None):
    with open(path, "rb") as f:
        magic = f.readline().strip()
        assert magic == b"PABT1", f"not a parsec_amd trace: {magic!r}"
        header = json.loads(f.readline())
        raw = f.read()
    rec = struct.Struct("<QQIHHQ")
    assert header["rec_bytes"] == rec.size
    classes = {int(k): v for k, v in header["classes"].items()}
    kinds = {int(k): v for k, v in header["kinds"].items()}
    events = []
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
            "pid": 1 if tid >= 1000 else 0,
            "tid": tid - 1000 if tid >= 1000 else tid,
            "args": {"seq": seq},
        })
    for lane in sorted(gpu_lanes):
        events.append({"name": "thread_name", "ph": "M", "pid": 1,
                       "tid": lane - 1000,
                       "args": {"name": f"gpu exec stream {lane - 1000}"}})
    if gpu_lanes:
        events.append({"name": "process_name", "ph": "M", "pid": 1,
                       "args": {"name": "GPU device spans"}})
    out = out_path or path + ".json"
    with open(out, "w") as f:
        json.dump({"traceEvents": events}, f)
    return out, len(events)


if __name__ == "__main__":
    out, n = convert(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
    print(f"wrote {n} events to {out}")
