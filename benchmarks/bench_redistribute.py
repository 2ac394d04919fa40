#!/usr/bin/env python3
"""Redistribution bandwidth (tests/collections/redistribute timing analog):
move an n x n fp64 matrix between two different block-cyclic layouts and
report effective bandwidth. Single rank measures the engine's tile-copy
path; two ranks measure real protocol transfers.

  RANK=0 WORLD_SIZE=2 python benchmarks/bench_redistribute.py &
  RANK=1 WORLD_SIZE=2 python benchmarks/bench_redistribute.py
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=8192)
    ap.add_argument("--tile", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()

    import parsec_amd as pm

    ctx = pm.init_distributed(nworkers=4, gpu=-2 if
                              pm.hip_device_count() == 0 else -1)
    w = ctx.world
    n, nb = args.size, args.tile
    S = pm.TiledMatrix(ctx, n, n, nb, nb, w, 1)
    D = pm.TiledMatrix(ctx, n, n, nb, nb, 1, w)  # transposed rank grid
    tp = pm.Dtd(ctx)
    pm.insert_full_fill(tp, S, 1)
    tp.wait()
    ctx.barrier()

    def run(iters):
        tp = pm.Dtd(ctx)
        for _ in range(iters):
            pm.insert_redistribute(tp, S, D)
            pm.insert_redistribute(tp, D, S)
        tp.wait()
        ctx.gpu_sync()
        ctx.barrier()

    run(1)
    t0 = time.perf_counter()
    run(args.iters)
    dt = time.perf_counter() - t0
    total = 2 * args.iters * n * n * 8
    if ctx.rank == 0:
        print(json.dumps({
            "metric": "redistribute bandwidth",
            "GB_s": round(total / dt / 1e9, 2),
            "n": n, "tile": nb, "world": w, "iters": args.iters,
        }), flush=True)
    del S, D, ctx


if __name__ == "__main__":
    main()
