"""Live runtime dashboard (tools/aggregator_visu analog).

A running context started with ``PARSEC_MCA_live_stats=<path>`` publishes a
JSON snapshot of its counters to ``<path>.<rank>`` every
``live_stats_interval_ms`` (atomic replace — never torn). This tool tails
those files and renders a per-rank table with task/byte rates computed from
consecutive snapshots.

Usage:
    python -m parsec_amd.tools.live_top <path> [--interval 1.0] [--once]
"""
import argparse
import glob
import json
import os
import sys
import time


def read_snapshots(prefix):
    snaps = {}
    for f in sorted(glob.glob(prefix + ".*")):
        if f.endswith(".tmp"):
            continue
        try:
            with open(f) as fh:
                s = json.load(fh)
            snaps[s.get("rank", f)] = s
        except (OSError, ValueError):
            continue  # mid-replace or gone: next tick
    return snaps


def render(snaps, prev, dt):
    rows = []
    hdr = (f"{'rank':>4} {'up(s)':>8} {'ready':>6} {'cpu':>10} {'gpu':>10} "
           f"{'tasks/s':>9} {'steals':>8} {'comm MB':>9} {'evict':>6}")
    rows.append(hdr)
    rows.append("-" * len(hdr))
    for r in sorted(snaps):
        s = snaps[r]
        done = s.get("tasks_cpu", 0) + s.get("tasks_gpu", 0)
        rate = 0.0
        if r in prev and dt > 0:
            p = prev[r]
            rate = (done - p.get("tasks_cpu", 0) - p.get("tasks_gpu", 0)) / dt
        rows.append(
            f"{s.get('rank', 0):>4} {s.get('uptime_s', 0):>8.1f} "
            f"{s.get('ready_queue', 0):>6} {s.get('tasks_cpu', 0):>10} "
            f"{s.get('tasks_gpu', 0):>10} {rate:>9.0f} "
            f"{s.get('steals', 0):>8} "
            f"{s.get('comm_bytes', 0) / 1e6:>9.1f} "
            f"{s.get('gpu_evictions', 0):>6}")
    return "\n".join(rows)


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("path", help="live_stats path prefix (without .<rank>)")
    ap.add_argument("--interval", type=float, default=1.0)
    ap.add_argument("--once", action="store_true",
                    help="print one snapshot table and exit")
    args = ap.parse_args(argv)
    prev, tprev = {}, time.time()
    while True:
        snaps = read_snapshots(args.path)
        now = time.time()
        if args.once:
            if not snaps:
                print(f"no live snapshots at {args.path}.*", file=sys.stderr)
                return 1
            print(render(snaps, prev, now - tprev))
            return 0
        os.system("clear" if os.name == "posix" else "cls")
        print(f"parsec_amd live_top — {args.path}.*  "
              f"({len(snaps)} rank(s), refresh {args.interval}s; ctrl-c quits)")
        print(render(snaps, prev, now - tprev) if snaps
              else "waiting for snapshots...")
        prev, tprev = snaps, now
        time.sleep(args.interval)


if __name__ == "__main__":
    sys.exit(main())
