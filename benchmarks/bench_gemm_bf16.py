#!/usr/bin/env python3
"""bf16 MFMA tile-GEMM DAG benchmark (BASELINE.json config 5).

C (fp32) = At^T B over bf16 tiles; whole-job TFLOP/s on the DTD engine.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=65536)
    ap.add_argument("--kdim", type=int, default=65536)
    ap.add_argument("--tile", type=int, default=8192)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    args = ap.parse_args()

    import parsec_amd as pm

    ctx = pm.init_distributed(nworkers=4)
    world = ctx.world
    p, q = (1, world)
    M = N = args.size
    K = args.kdim
    kb = mb = args.tile
    At = pm.TiledMatrix(ctx, K, M, kb, mb, p, q, elem_size=2)
    B = pm.TiledMatrix(ctx, K, N, kb, mb, p, q, elem_size=2)
    C = pm.TiledMatrix(ctx, M, N, mb, mb, p, q, elem_size=4)

    def step(n):
        tp = pm.Dtd(ctx)
        for _ in range(n):
            pm.insert_fill_bf16(tp, At, 1)
            pm.insert_fill_bf16(tp, B, 2)
            pm.insert_gemm_bf16(tp, At, B, C)
        tp.wait()

    step(args.warmup)
    ctx.gpu_sync()
    ctx.barrier()
    t0 = time.perf_counter()
    step(args.steps)
    ctx.gpu_sync()
    ctx.barrier()
    dt = time.perf_counter() - t0
    tf = args.steps * 2.0 * M * N * K / dt / 1e12
    if ctx.rank == 0:
        print(json.dumps({
            "metric": "TFLOP/s bf16 tile-GEMM DAG", "value": round(tf, 1),
            "n_gpus": world, "M": M, "N": N, "K": K, "tile": args.tile,
            "dtype": "bf16(fp32 acc)", "data": "synthetic",
        }), flush=True)
    del At, B, C, ctx


if __name__ == "__main__":
    main()
