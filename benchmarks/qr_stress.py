#!/usr/bin/env python3
"""QR correctness stress: repeat small hand-panel QR, report failures."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import parsec_amd as pm

iters = int(sys.argv[1]) if len(sys.argv) > 1 else 20
ctx = pm.Context(nworkers=2, rank=0, world=1)
n, nb = 1024, 256
fails = 0
for it in range(iters):
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    if os.environ.get("QR_STRESS_FILL", "host") == "host":
        rng = np.random.default_rng(42 + it)
        for tm in range(A.mt):
            for tn in range(A.nt):
                A.tile_numpy_set(tm, tn, rng.standard_normal((nb, nb)))
    else:
        tp0 = pm.Dtd(ctx)
        pm.insert_full_fill(tp0, A, 42 + it)
        tp0.wait()
    A0 = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(A.nt):
            A0[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
    tp = pm.Dtd(ctx)
    pm.insert_geqrf(tp, A)
    tp.wait()
    R = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(A.nt):
            R[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
    R = np.triu(R)
    err = np.abs(R.T @ R - A0.T @ A0).max() / np.abs(A0.T @ A0).max()
    ok = err < 1e-12
    if not ok:
        fails += 1
        print(f"iter {it}: FAIL err={err:.3e}", flush=True)
        # forensics: sign-normalized comparison against numpy R, per tile
        Rnp = np.linalg.qr(A0, mode="r")
        sgn = np.sign(np.diag(Rnp)) * np.sign(np.diag(R) + (np.diag(R) == 0))
        Rn = Rnp * sgn[:, None]
        for tm in range(A.mt):
            row = []
            for tn in range(A.nt):
                d = np.abs(R[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] -
                           Rn[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb]).max()
                row.append(f"{d:9.2e}")
            print("   tile row", tm, " ".join(row), flush=True)
        # column-resolved: first column where R diverges
        cd = np.abs(R - Rn).max(axis=0)
        bad = np.nonzero(cd > 1e-8)[0]
        print(f"   first bad col {bad[0] if len(bad) else -1}, nbad={len(bad)}",
              flush=True)
    del A
print(f"fails={fails}/{iters}", flush=True)
del ctx
