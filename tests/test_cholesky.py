"""Numerics: tiled DTD Cholesky vs NumPy fp64 reference (CPU chores)."""
import numpy as np
import pytest

import parsec_amd as pm


def assemble_lower(A, n, nb):
    M = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(min(tm + 1, A.nt)):
            r, c = A.tile_rows(tm), A.tile_cols(tn)
            M[tm * nb:tm * nb + r, tn * nb:tn * nb + c] = A.tile_numpy(tm, tn)
    return M


@pytest.mark.parametrize("n,nb", [(256, 64), (200, 64), (192, 48)])
def test_cholesky_vs_numpy(ctx, n, nb):
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    tp = pm.Dtd(ctx)
    pm.insert_spd_fill(tp, A, 42)
    tp.wait()
    M = assemble_lower(A, n, nb)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)

    tp2 = pm.Dtd(ctx)
    pm.insert_potrf(tp2, A)
    tp2.wait()
    L = np.tril(assemble_lower(A, n, nb))
    err = np.abs(L - L0).max()
    assert err < 1e-10, f"max err {err}"


def test_fill_deterministic(ctx):
    A = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
    tp = pm.Dtd(ctx)
    pm.insert_spd_fill(tp, A, 7)
    tp.wait()
    t1 = A.tile_numpy(1, 0)
    tp2 = pm.Dtd(ctx)
    pm.insert_spd_fill(tp2, A, 7)
    tp2.wait()
    t2 = A.tile_numpy(1, 0)
    assert np.array_equal(t1, t2)
    # symmetric: tile(1,0) vs transpose region of a symmetric fill
    B = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
    tp3 = pm.Dtd(ctx)
    # fill full matrix (including upper) through direct tile fills
    pm.insert_spd_fill(tp3, B, 7)
    tp3.wait()
    d = B.tile_numpy(0, 0)
    assert np.array_equal(d, d.T)  # diagonal tile symmetric


def test_panel_cholesky_vs_numpy(ctx):
    """Panel-granularity variant (bench --algo panel) vs NumPy."""
    n, nb = 256, 64
    A = pm.TiledMatrix(ctx, n, n, n, nb, 1, 1)
    tp = pm.Dtd(ctx)
    pm.insert_panel_fill(tp, A, 42)
    tp.wait()
    M = np.zeros((n, n))
    for k in range(A.nt):
        M[:, k * nb:(k + 1) * nb] = A.tile_numpy(0, k)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)
    tp2 = pm.Dtd(ctx)
    pm.insert_potrf_panel(tp2, A)
    tp2.wait()
    L = np.zeros((n, n))
    for k in range(A.nt):
        L[:, k * nb:(k + 1) * nb] = A.tile_numpy(0, k)
    err = np.abs(np.tril(L) - L0).max()
    assert err < 1e-10, f"panel cholesky max err {err}"


def _full(M, lower=False):
    import numpy as np
    out = np.zeros((M.m, M.n))
    for i in range(M.mt):
        for j in range(min(i + 1, M.nt) if lower else M.nt):
            out[i * M.mb:i * M.mb + M.tile_rows(i),
                j * M.nb:j * M.nb + M.tile_cols(j)] = M.tile_numpy(i, j)
    return out


def test_posv_vs_numpy(ctx):
    """insert_posv (dplasma dposv analog): factor + two triangular solve
    sweeps against numpy.linalg.solve, partial RHS tiles included."""
    import numpy as np
    n, nb, nrhs = 320, 64, 96
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1, sym=True)
    B = pm.TiledMatrix(ctx, n, nrhs, nb, nb, 1, 1)
    tp = pm.Dtd(ctx)
    pm.insert_spd_fill(tp, A, 11)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()
    Lo = np.tril(_full(A, lower=True))
    Af = Lo + np.tril(Lo, -1).T
    Bf = _full(B)
    tp2 = pm.Dtd(ctx)
    pm.insert_posv(tp2, A, B)
    tp2.wait()
    ref = np.linalg.solve(Af, Bf)
    err = abs(_full(B) - ref).max() / abs(ref).max()
    assert err < 1e-11, err
