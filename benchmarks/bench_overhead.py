#!/usr/bin/env python3
"""Runtime overhead microbenchmark: tasks/second through the full
insert->chain->schedule->execute->complete path.

The reference quotes per-task overheads in the us range (DTD docs,
dtd.md performance checklist); this prints the equivalent numbers here:
  - empty C++ tasks on one tile chain (serial dependency, measures the
    critical-path overhead per task)
  - empty C++ tasks on many tiles (parallel, measures throughput with
    all workers stealing)
Run on CPU or GPU boxes (tasks are CPU no-ops either way).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tasks", type=int, default=200000)
    ap.add_argument("--workers", type=int, default=4)
    args = ap.parse_args()

    import parsec_amd as pm

    ctx = pm.Context(nworkers=args.workers, rank=0, world=1, gpu=-2)
    ntiles = 256
    A = pm.TiledMatrix(ctx, ntiles * 8, 8, 8, 8, 1, 1)

    # noop task class: insert_apply_scale with alpha=1 beta=0 is the
    # cheapest C++ task (one tile INOUT, 64-element scale)
    def run(serial, n):
        tp = pm.Dtd(ctx)
        if serial:
            for _ in range(n):
                pm.insert_apply_scale(tp, Aser, 1.0, 0.0)
        else:
            per = n // ntiles
            for _ in range(per):
                pm.insert_apply_scale(tp, A, 1.0, 0.0)
        tp.wait()

    Aser = pm.TiledMatrix(ctx, 8, 8, 8, 8, 1, 1)
    tp0 = pm.Dtd(ctx)
    pm.insert_full_fill(tp0, A, 1)
    pm.insert_full_fill(tp0, Aser, 1)
    tp0.wait()
    run(True, 1000)  # warmup
    n_serial = min(args.tasks // 4, 50000)
    t0 = time.perf_counter()
    run(True, n_serial)
    dt_serial = time.perf_counter() - t0

    run(False, ntiles * 20)  # warmup
    t0 = time.perf_counter()
    run(False, args.tasks)
    dt_par = time.perf_counter() - t0

    print(json.dumps({
        "metric": "runtime task overhead",
        "serial_us_per_task": round(dt_serial / n_serial * 1e6, 3),
        "parallel_tasks_per_sec": round(args.tasks / dt_par, 0),
        "parallel_us_per_task_per_worker":
            round(dt_par / args.tasks * 1e6 * args.workers, 3),
        "workers": args.workers,
        "tasks": args.tasks,
    }), flush=True)
    del A, Aser, ctx


if __name__ == "__main__":
    main()
