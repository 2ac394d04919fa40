"""Distributed task-dataflow merge sort (tests/apps/merge_sort analog).

A binary merge tree over an IrregularCollection: leaves hold chunks of the
input (distributed round-robin), internal nodes hold merged runs of
doubling size. Each merge task reads its two children and writes the
parent — the runtime derives all ordering and inter-rank movement.

Run:  RANK=0 WORLD_SIZE=1 python examples/merge_sort.py [n]
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import parsec_amd as pm  # noqa: E402


def key(level, idx):
    return (level << 32) | idx


def merge_sort(ctx, values, chunk=1 << 12):
    """Sort `values` (same array on every rank, SPMD) via the task tree."""
    n = len(values)
    nleaf = (n + chunk - 1) // chunk
    levels = max(1, (nleaf - 1).bit_length() + 1)
    nodes = pm.IrregularCollection(ctx)

    tp = pm.Dtd(ctx, "msort")
    # leaves: fill + local sort
    for i in range(nleaf):
        lo, hi = i * chunk, min(n, (i + 1) * chunk)
        d = nodes.add(key(0, i), i % ctx.world, (hi - lo) * 8)

        def fill(buf, lo=lo, hi=hi):
            a = np.frombuffer(buf, dtype=np.float64)
            a[:] = np.sort(values[lo:hi])

        tp.insert_py(fill, flows=[(d, pm.ACCESS_OUT)], with_data=True)

    # merge tree
    width = nleaf
    for lv in range(1, levels):
        nw = (width + 1) // 2
        if width == 1:
            break
        for i in range(nw):
            li, ri = 2 * i, 2 * i + 1
            left = nodes.at(key(lv - 1, li))
            if ri < width:
                right = nodes.at(key(lv - 1, ri))
                out = nodes.add(key(lv, i), i % ctx.world,
                                left.nbytes + right.nbytes)

                def merge(lbuf, rbuf, obuf):
                    a = np.frombuffer(lbuf, dtype=np.float64)
                    b = np.frombuffer(rbuf, dtype=np.float64)
                    o = np.frombuffer(obuf, dtype=np.float64)
                    o[:] = np.concatenate([a, b])
                    o.sort(kind="mergesort")

                tp.insert_py(merge, flows=[(left, pm.ACCESS_IN),
                                           (right, pm.ACCESS_IN),
                                           (out, pm.ACCESS_OUT)],
                             with_data=True)
            else:  # odd node promotes
                out = nodes.add(key(lv, i), i % ctx.world, left.nbytes)

                def promote(lbuf, obuf):
                    np.frombuffer(obuf, dtype=np.float64)[:] = \
                        np.frombuffer(lbuf, dtype=np.float64)

                tp.insert_py(promote, flows=[(left, pm.ACCESS_IN),
                                             (out, pm.ACCESS_OUT)],
                             with_data=True)
        width = nw
    tp.wait()
    root_key = key(levels - 1, 0) if nleaf > 1 else key(0, 0)
    return nodes, root_key


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 1 << 16
    ctx = pm.Context(gpu=-2)
    rng = np.random.default_rng(7)
    values = rng.standard_normal(n)  # same seed on every rank (SPMD)
    nodes, root_key = merge_sort(ctx, values)
    if nodes.at(root_key).home_rank == ctx.rank:
        out = np.frombuffer(nodes.bytes_get(root_key), dtype=np.float64)
        assert np.array_equal(out, np.sort(values))
        print(f"sorted {n} values on {ctx.world} rank(s)")
    ctx.barrier()


if __name__ == "__main__":
    main()
