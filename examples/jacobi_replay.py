"""Jacobi-style fixed-point iteration with hipGraph replay.

The steady-state inner loop of an iterative solver is the textbook
launch-bound workload: the SAME small-tile DAG every sweep. This example
builds the affine iteration

    x_{k+1} = M x_k + b      (rho(M) < 1  =>  x* = (I - M)^{-1} b)

as a tiled DAG and, on a GPU, captures ONE double sweep (y = M^x, x = M^y
on the augmented system M^ = [[M, b], [0, 1]], x^ = [x; 1] — augmentation
keeps the iteration a pure tiled GEMM DAG with no per-sweep host work)
into a hipGraph with `tp.capture_begin()` / `tp.capture_end()`. Replays
then cost one hipGraphLaunch per double sweep. On a CPU-only container the
same pools re-run through the engine (same numerics, no graph).

Run:  python examples/jacobi_replay.py [n_tiles] [nb] [sweeps]
"""
import sys
import time

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])
import parsec_amd as pm  # noqa: E402


def main(nt=4, nb=64, sweeps=50):
    n = nt * nb
    rng = np.random.default_rng(7)
    # rho(M) < 1: random contraction + the augmentation row [0 ... 0 1]
    M = rng.standard_normal((n, n))
    M *= 0.6 / np.abs(np.linalg.eigvals(M)).max()
    b = rng.standard_normal(n)
    Mh = np.zeros((n + nb, n + nb))
    Mh[:n, :n] = M
    Mh[:n, n] = b
    Mh[n, n] = 1.0
    xh = np.zeros(n + nb)
    xh[n] = 1.0

    ctx = pm.Context(nworkers=4)
    A = pm.TiledMatrix(ctx, n + nb, n + nb, nb, nb, 1, 1)
    X = pm.TiledMatrix(ctx, n + nb, 1, nb, 1, 1, 1)
    Y = pm.TiledMatrix(ctx, n + nb, 1, nb, 1, 1, 1)
    for i in range(A.mt):
        for j in range(A.nt):
            A.tile_numpy_set(i, j, Mh[i * nb:(i + 1) * nb,
                                      j * nb:(j + 1) * nb])
        X.tile_numpy_set(i, 0, xh[i * nb:(i + 1) * nb].reshape(-1, 1))

    def double_sweep(tp):
        pm.insert_gemm_fp64(tp, A, X, Y)  # y = M^ x
        pm.insert_gemm_fp64(tp, A, Y, X)  # x = M^ y

    t0 = time.perf_counter()
    g = None
    if ctx.has_gpu:
        tp = pm.Dtd(ctx)
        tp.capture_begin()
        double_sweep(tp)
        g = tp.capture_end()          # record pass = first double sweep
        g.launch(sweeps - 1)          # the steady-state loop: one
        ctx.gpu_sync()                # hipGraphLaunch per double sweep
        mode = f"hipGraph replay ({g.n_tasks} tasks -> {g.nodes} nodes)"
    else:
        for _ in range(sweeps):
            tp = pm.Dtd(ctx)
            double_sweep(tp)
            tp.wait()
        mode = "pool re-run (no GPU)"
    dt = (time.perf_counter() - t0) / sweeps * 1e3

    got = np.concatenate([X.tile_numpy(i, 0).ravel() for i in range(A.mt)])
    xstar = np.linalg.solve(np.eye(n) - M, b)
    err = np.abs(got[:n] - xstar).max() / np.abs(xstar).max()
    print(f"jacobi_replay: {mode}; {sweeps} double sweeps, "
          f"{dt:.3f} ms/sweep, fixed-point err {err:.2e}")
    assert err < 1e-8, err
    # HIP-graph lifetime rule: destroy the graph BEFORE the collections it
    # pins (same as destroying buffers under a live cudaGraph is UB)
    del g
    del A, X, Y, ctx
    return err


if __name__ == "__main__":
    args = [int(a) for a in sys.argv[1:4]]
    main(*args)
