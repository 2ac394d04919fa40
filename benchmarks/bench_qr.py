#!/usr/bin/env python3
"""Tile QR (dgeqrf) benchmark: whole-job GFLOP/s on the DTD engine.

Current status (profiles/RESULTS.md): numerics-complete; throughput is
bounded by rocSOLVER's unblocked panel kernels — the hand panel path
(PARSEC_MCA_chore_qr=hand) is the round-2 optimization target.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=16384)
    ap.add_argument("--tile", type=int, default=1024)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--algo", type=str,
                    default=os.environ.get("PARSEC_MCA_qr_algo", "house"),
                    choices=["house", "bcgs"])
    args = ap.parse_args()

    import parsec_amd as pm

    has_gpu = pm.hip_device_count() > 0
    n, nb = (args.size, args.tile) if has_gpu else (512, 128)
    ctx = pm.init_distributed(nworkers=4)
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    R = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1) if args.algo == "bcgs"         else None

    def run(nsteps):
        tp = pm.Dtd(ctx)
        for _ in range(nsteps):
            pm.insert_full_fill(tp, A, 3)
            if args.algo == "bcgs":
                pm.insert_geqrf_bcgs(tp, A, R)
            else:
                pm.insert_geqrf(tp, A)
        tp.wait()

    if args.warmup:
        run(args.warmup)
    ctx.gpu_sync()
    t0 = time.perf_counter()
    run(args.steps)
    ctx.gpu_sync()
    dt = time.perf_counter() - t0
    flops = 4.0 / 3.0 * n**3  # dgeqrf square
    print(json.dumps({
        "metric": "GFLOP/s tile QR dgeqrf",
        "value": round(args.steps * flops / dt / 1e9, 1),
        "unit": "GFLOP/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt / args.steps * 1e3, 2),
        "higher_is_better": True,
        "dtype": "fp64",
        "data": "synthetic",
        "config": {"model": "tile_qr_dgeqrf", "N": n, "tile": nb,
                   "algo": args.algo,
                   "chore_qr": os.environ.get("PARSEC_MCA_chore_qr", "hand")},
    }), flush=True)
    del A, ctx


if __name__ == "__main__":
    main()
