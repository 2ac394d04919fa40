#!/usr/bin/env python3
"""PTG materialization ceiling: insertion rate + memory at large instance
counts (the reference's jdf2c never materializes the instance space,
jdf2c.c:3047-3454 — this documents what the materializing compiler costs
and where it stops being free).

Runs a 1-flow chain taskpool of --tasks instances with empty CPU bodies,
reports insertion+drain wall time and peak RSS.
"""
import argparse
import os
import resource
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

JDF = """
A  [ type="parsec_data_collection_t*" ]
NT [ type="int" ]

Step(k)

k = 0 .. NT-1

: A( k %% 64, 0 )

RW X <- (k > 63) ? X Step(k - 64) : A(k %% 64, 0)
     -> (k + 64 < NT) ? X Step(k + 64)
     -> A(k %% 64, 0)

BODY
{
}
END
"""


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tasks", type=int, default=1000000)
    ap.add_argument("--compact", action="store_true",
                    help="never-materialized iteration (jdf2c analog)")
    args = ap.parse_args()
    import tempfile
    import json
    import parsec_amd as pm
    from parsec_amd.ptg import compile_jdf

    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "scale.jdf")
        with open(p, "w") as f:
            f.write(JDF.replace("%%", "%"))
        mod = compile_jdf(p)
        ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
        A = pm.TiledMatrix(ctx, 64, 1, 1, 1, 1, 1)
        import struct
        for k in range(64):
            A.tile_bytes_set(k, 0, struct.pack("<q", 0))
        rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        t0 = time.time()
        tp = pm.Dtd(ctx, "scale")
        mod.build(ctx, tp, compact=args.compact, A=A, NT=args.tasks)
        t_build = time.time() - t0
        tp.wait()
        t_total = time.time() - t0
        rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        print(json.dumps({
            "metric": ("PTG compact iteration" if args.compact else "PTG materialized insertion"),
            "tasks": args.tasks,
            "build_s": round(t_build, 3),
            "total_s": round(t_total, 3),
            "tasks_per_s": round(args.tasks / t_total),
            "peak_rss_mb": round(rss1 / 1024),
            "delta_rss_mb": round((rss1 - rss0) / 1024),
            "bytes_per_task": round((rss1 - rss0) * 1024 / args.tasks),
        }), flush=True)
        del A, ctx


if __name__ == "__main__":
    main()
