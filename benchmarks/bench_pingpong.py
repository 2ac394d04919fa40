#!/usr/bin/env python3
"""Comm-engine ping-pong: RTT and bandwidth through the dataflow protocol
(tests/apps/pingpong rtt.jdf + bandwidth.jdf analog).

A single tile bounces rank0 -> rank1 -> rank0 ... via alternating INOUT
tasks; every hop is a real protocol transfer (send/recv tasks, channel
sequencing). Run with two ranks:

  RANK=0 WORLD_SIZE=2 python benchmarks/bench_pingpong.py &
  RANK=1 WORLD_SIZE=2 python benchmarks/bench_pingpong.py

Prints per-size RTT (us) and bandwidth (MB/s) from rank 0. Works on the
TCP engine (CPU) and, on a multi-GPU node, over RCCL/xGMI with
PARSEC_MCA_comm_kind=rccl (device-resident payloads).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--hops", type=int, default=200)
    ap.add_argument("--sizes", type=str, default="4096,262144,16777216")
    args = ap.parse_args()

    import parsec_amd as pm

    ctx = pm.init_distributed(nworkers=2, gpu=-2 if
                              pm.hip_device_count() == 0 else -1)
    assert ctx.world == 2, "ping-pong needs exactly 2 ranks"
    rank = ctx.rank
    results = []
    for size in (int(s) for s in args.sizes.split(",")):
        nb = max(1, int((size // 8) ** 0.5))
        A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 2, 1)  # tile(0,0) on rank 0
        tp = pm.Dtd(ctx, f"pp{size}")
        pm.insert_full_fill(tp, A, 1)
        tp.wait()
        ctx.barrier()

        def run(hops):
            tp = pm.Dtd(ctx, "pp")
            t = A.tile(0, 0)
            for h in range(hops):
                # INOUT on alternating ranks: every hop moves the payload
                tp.insert_py(lambda: None, flows=[(t, pm.ACCESS_INOUT)],
                             rank=h % 2)
            tp.wait()
            ctx.barrier()

        run(10)  # warmup
        t0 = time.perf_counter()
        run(args.hops)
        dt = time.perf_counter() - t0
        bytes_moved = nb * nb * 8
        rtt_us = dt / args.hops * 2 * 1e6  # two hops = one round trip
        bw = bytes_moved * args.hops / dt / 1e6
        results.append({"bytes": bytes_moved, "rtt_us": round(rtt_us, 1),
                        "MB_s": round(bw, 1)})
        del A
    if rank == 0:
        print(json.dumps({"metric": "comm ping-pong", "hops": args.hops,
                          "engine": "rccl" if pm.hip_device_count() else "tcp",
                          "results": results}), flush=True)
    del ctx


if __name__ == "__main__":
    main()
