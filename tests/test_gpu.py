"""GPU tests (MI355X): HIP engine numerics vs NumPy fp64, chore variants.

Numerics tests compare the full GPU path (rocBLAS/rocSOLVER chores and the
hand-written CDNA4 MFMA chores in src/kernels_hip.cpp) against a plain fp64
NumPy reference on the same synthetic input.
"""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

import parsec_amd as pm

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gctx():
    c = pm.Context(nworkers=2, rank=0, world=1)
    assert c.has_gpu
    yield c
    del c


def assemble_lower(A, n, nb):
    M = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(min(tm + 1, A.nt)):
            r, c = A.tile_rows(tm), A.tile_cols(tn)
            M[tm * nb:tm * nb + r, tn * nb:tn * nb + c] = A.tile_numpy(tm, tn)
    return M


def test_gpu_cholesky_numerics(gctx):
    n, nb = 2048, 256
    A = pm.TiledMatrix(gctx, n, n, nb, nb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_spd_fill(tp, A, 42)
    tp.wait()
    M = assemble_lower(A, n, nb)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)
    tp2 = pm.Dtd(gctx)
    pm.insert_potrf(tp2, A)
    tp2.wait()
    L = np.tril(assemble_lower(A, n, nb))
    err = np.abs(L - L0).max()
    assert err < 1e-8, f"max err {err}"
    assert gctx.gpu_stats()["tasks"] > 0, "GPU engine did not execute tasks"


def test_gpu_partial_edge_tiles(gctx):
    n, nb = 1000, 192  # non-dividing tile size exercises edge-tile args
    A = pm.TiledMatrix(gctx, n, n, nb, nb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_spd_fill(tp, A, 3)
    tp.wait()
    M = assemble_lower(A, n, nb)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)
    tp2 = pm.Dtd(gctx)
    pm.insert_potrf(tp2, A)
    tp2.wait()
    L = np.tril(assemble_lower(A, n, nb))
    err = np.abs(L - L0).max()
    assert err < 1e-8, f"max err {err}"


def _run_chol_subprocess(chore, n=2048, nb=512):
    """Run a Cholesky + residual check in a subprocess with a chore env."""
    code = f"""
import numpy as np, sys
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
ctx = pm.Context(nworkers=2, rank=0, world=1)
n, nb = {n}, {nb}
A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 42); tp.wait()
M = np.zeros((n,n))
for tm in range(A.mt):
    for tn in range(tm+1):
        M[tm*nb:tm*nb+A.tile_rows(tm), tn*nb:tn*nb+A.tile_cols(tn)] = A.tile_numpy(tm,tn)
M = np.tril(M) + np.tril(M,-1).T
L0 = np.linalg.cholesky(M)
tp2 = pm.Dtd(ctx); pm.insert_potrf(tp2, A); tp2.wait()
L = np.zeros((n,n))
for tm in range(A.mt):
    for tn in range(tm+1):
        L[tm*nb:tm*nb+A.tile_rows(tm), tn*nb:tn*nb+A.tile_cols(tn)] = A.tile_numpy(tm,tn)
err = np.abs(np.tril(L)-L0).max()
print("ERR", err)
assert err < 1e-8, err
del A, ctx
"""
    env = dict(os.environ)
    env["PARSEC_MCA_chore_gemm"] = chore
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    return r.stdout


def test_hip_mfma_gemm_chore_numerics():
    """Hand-written fp64 MFMA dgemm chore vs NumPy (and vs rocBLAS path)."""
    _run_chol_subprocess("hip")


def test_rocblas_chore_numerics():
    _run_chol_subprocess("rocblas")


def test_bench_contract():
    """bench.py emits the JSON contract line and a sane value."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--matrix-size",
         "8192", "--tile", "1024", "--steps", "2", "--warmup", "1"],
        env=env, capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 1
    assert out["value"] > 0
    assert out["dtype"] == "fp64"


def test_eviction_under_memory_pressure():
    """Force LRU eviction+writeback with a hard HBM cap smaller than the
    working set; the factorization must still be numerically correct."""
    code = f"""
import numpy as np, sys
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("gpu_mem_limit_mb", "96")  # matrix alone is ~134 MB
pm.param_set("gpu_max_inflight", "4")
ctx = pm.Context(nworkers=2, rank=0, world=1)
n, nb = 4096, 512
A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 42); tp.wait()
M = np.zeros((n,n))
for tm in range(A.mt):
    for tn in range(tm+1):
        M[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm,tn)
M = np.tril(M) + np.tril(M,-1).T
L0 = np.linalg.cholesky(M)
tp2 = pm.Dtd(ctx); pm.insert_potrf(tp2, A); tp2.wait()
L = np.zeros((n,n))
for tm in range(A.mt):
    for tn in range(tm+1):
        L[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm,tn)
err = np.abs(np.tril(L)-L0).max()
print("EVICT_ERR", err)
assert err < 1e-8, err
del A, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr


def test_gpu_random_dag_vs_oracle():
    """Fuzz THROUGH the GPU engine: random scale/add/copy DAGs on GPU
    chores (H2D staging, fences, eviction interplay) vs numpy oracle."""
    import numpy as np
    import random
    ctx = pm.Context(nworkers=2, rank=0, world=1)
    assert ctx.has_gpu
    rng = random.Random(777)
    nb, NT = 256, 8
    mats = [pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1) for _ in range(NT)]
    ref = []
    tp = pm.Dtd(ctx)
    for i, M in enumerate(mats):
        v = np.full((nb, nb), float(i + 1))
        M.tile_numpy_set(0, 0, v)
        ref.append(v.copy())
    for _ in range(300):
        op = rng.choice(["scale", "add", "copy"])
        if op == "scale":
            i = rng.randrange(NT)
            a, b = rng.uniform(0.5, 1.5), rng.uniform(-1, 1)
            pm.insert_apply_scale(tp, mats[i], a, b)
            ref[i] = ref[i] * a + b
        elif op == "add":
            i, j = rng.randrange(NT), rng.randrange(NT)
            if i == j:
                continue
            pm.insert_reduce_sum(tp, mats[i], mats[j])
            ref[j] = ref[j] + ref[i]
        else:
            i, j = rng.randrange(NT), rng.randrange(NT)
            if i == j:
                continue
            pm.insert_redistribute(tp, mats[i], mats[j])
            ref[j] = ref[i].copy()
    tp.wait()
    assert ctx.gpu_stats()["tasks"] > 200  # the GPU actually ran the DAG
    for i, M in enumerate(mats):
        got = M.tile_numpy(0, 0)
        assert np.allclose(got, ref[i]), f"tile {i} diverged"
    del mats, ctx
