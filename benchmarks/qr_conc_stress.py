#!/usr/bin/env python3
"""Concurrency amplifier: many INDEPENDENT 256^2 GEQRT tiles at once."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import parsec_amd as pm

iters = int(sys.argv[1]) if len(sys.argv) > 1 else 5
NT = int(os.environ.get("QR_NT", "16"))
ctx = pm.Context(nworkers=2, rank=0, world=1)
nb = 256
fails = 0
for it in range(iters):
    A = pm.TiledMatrix(ctx, nb * NT, nb, nb, nb, 1, 1)
    rng = np.random.default_rng(100 + it)
    srcs = []
    for tm in range(NT):
        v = rng.standard_normal((nb, nb))
        srcs.append(v)
        A.tile_numpy_set(tm, 0, v)
    mats = []
    tp = pm.Dtd(ctx)
    for tm in range(NT):
        M = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
        M.tile_numpy_set(0, 0, srcs[tm])
        mats.append(M)
        pm.insert_geqrf(tp, M)
    tp.wait()
    for tm in range(NT):
        R = np.triu(mats[tm].tile_numpy(0, 0))
        A0 = srcs[tm]
        err = np.abs(R.T @ R - A0.T @ A0).max() / np.abs(A0.T @ A0).max()
        if err > 1e-12:
            fails += 1
            print(f"iter {it} tile {tm}: FAIL err={err:.3e}", flush=True)
    del mats, A
print(f"fails={fails}", flush=True)
del ctx
