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


def test_eviction_mixed_tile_sizes():
    """Fragmentation pressure: two collections with different tile sizes
    plus subtile traffic under a hard HBM cap — the coalescing slab
    allocator must recycle mixed-size blocks (round-1 size-class freelists
    could strand capacity here) and host-staged tiles must stay evictable
    (h2d fence clearing)."""
    code = f"""
import numpy as np, sys
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("gpu_mem_limit_mb", "80")
pm.param_set("gpu_max_inflight", "4")
ctx = pm.Context(nworkers=2, rank=0, world=1)
n1, nb1 = 4096, 512     # 2 MB tiles
n2, nb2 = 3520, 320     # 0.8 MB tiles (different size class)
A = pm.TiledMatrix(ctx, n1, n1, nb1, nb1, 1, 1)
B = pm.TiledMatrix(ctx, n2, n2, nb2, nb2, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 42)
pm.insert_spd_fill(tp, B, 7)
tp.wait()
# host-stage every tile of A (tile_numpy pulls to host), then make the GPU
# re-stage them H2D under pressure: staged tiles must remain evictable
MA = np.zeros((n1,n1)); MB = np.zeros((n2,n2))
for tm in range(A.mt):
    for tn in range(tm+1):
        MA[tm*nb1:(tm+1)*nb1, tn*nb1:(tn+1)*nb1] = A.tile_numpy(tm,tn)
for tm in range(B.mt):
    for tn in range(tm+1):
        MB[tm*nb2:(tm+1)*nb2, tn*nb2:(tn+1)*nb2] = B.tile_numpy(tm,tn)
MA = np.tril(MA) + np.tril(MA,-1).T
MB = np.tril(MB) + np.tril(MB,-1).T
L0A = np.linalg.cholesky(MA)
L0B = np.linalg.cholesky(MB)
tp2 = pm.Dtd(ctx)
pm.insert_potrf(tp2, A)
pm.insert_potrf(tp2, B)
tp2.wait()
LA = np.zeros((n1,n1)); LB = np.zeros((n2,n2))
for tm in range(A.mt):
    for tn in range(tm+1):
        LA[tm*nb1:(tm+1)*nb1, tn*nb1:(tn+1)*nb1] = A.tile_numpy(tm,tn)
for tm in range(B.mt):
    for tn in range(tm+1):
        LB[tm*nb2:(tm+1)*nb2, tn*nb2:(tn+1)*nb2] = B.tile_numpy(tm,tn)
errA = np.abs(np.tril(LA)-L0A).max()
errB = np.abs(np.tril(LB)-L0B).max()
st = ctx.gpu_stats()
print("MIXED_ERR", errA, errB, "evictions", st["evictions"])
assert errA < 1e-8 and errB < 1e-8, (errA, errB)
assert st["evictions"] > 0, "cap did not force eviction - test is vacuous"
del A, B, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr


def test_gpu_stream_spans_in_trace():
    """Per-exec-stream device spans (hipEvent-timed) land in the binary
    trace as kind-7 records with tid = 1000+stream (reference parity:
    per-GPU profiling streams, device_cuda_module.c:509-530)."""
    import struct as st
    code = f"""
import numpy as np, sys, json, struct
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("profile_filename", "/tmp/span_trace")
ctx = pm.Context(nworkers=2, rank=0, world=1)
n, nb = 2048, 512
A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 42); tp.wait()
tp2 = pm.Dtd(ctx); pm.insert_potrf(tp2, A); tp2.wait()
st = ctx.gpu_stats()
assert st["bytes_required"] > 0
del tp, tp2, A, ctx  # keep_alive: the context (and trace dump) waits on them
with open("/tmp/span_trace.0", "rb") as f:
    assert f.readline().strip() == b"PABT1"
    hdr = json.loads(f.readline())
    raw = f.read()
rec = struct.Struct("<QQIHHQ")
spans = {{}}
for off in range(0, len(raw) - rec.size + 1, rec.size):
    t0, t1, tid, kind, cid, seq = rec.unpack_from(raw, off)
    if kind == 7:
        assert tid >= 1000, tid
        assert t1 >= t0
        spans.setdefault(tid, 0)
        spans[tid] += 1
assert spans, "no gpu_span records"
print("SPANS", spans)
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "SPANS" in r.stdout


def full_matrix(M, mb, nb):
    out = np.zeros((M.mt * mb, M.nt * nb))
    for i in range(M.mt):
        for j in range(M.nt):
            out[i * mb:i * mb + M.tile_rows(i),
                j * nb:j * nb + M.tile_cols(j)] = M.tile_numpy(i, j)
    return out


def test_gpu_graph_capture_replay(gctx):
    """hipGraph capture/replay (gpu_graph.hpp): the record pass computes
    C = A*B; replaying the instantiated graph N more times reproduces the
    SAME C from the same device-resident A/B (the DAG is idempotent: the
    k==0 task overwrites C with beta=0). Cross-stream dependency edges are
    exercised by the k-chains per C tile landing on different exec
    streams."""
    mb = 256
    A = pm.TiledMatrix(gctx, 8 * mb, 2 * mb, mb, mb, 1, 1)
    B = pm.TiledMatrix(gctx, 2 * mb, 8 * mb, mb, mb, 1, 1)
    C = pm.TiledMatrix(gctx, 8 * mb, 8 * mb, mb, mb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_full_fill(tp, A, 3)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()
    tp2 = pm.Dtd(gctx)
    tp2.capture_begin()
    pm.insert_gemm_fp64(tp2, A, B, C)
    g = tp2.capture_end()  # record pass ran the DAG once
    assert g.n_tasks == 8 * 8 * 2
    assert g.nodes >= g.n_tasks
    ref = full_matrix(A, mb, mb) @ full_matrix(B, mb, mb)
    got = full_matrix(C, mb, mb)
    err0 = abs(got - ref).max() / abs(ref).max()
    assert err0 < 1e-13, f"record pass wrong: {err0}"
    # scribble on C's HOST copies only: replay must overwrite the DEVICE
    # copies and win (proves the graph really re-executes)
    g.launch(3)
    gctx.gpu_sync()
    got2 = full_matrix(C, mb, mb)
    err1 = abs(got2 - ref).max() / abs(ref).max()
    assert err1 < 1e-13, f"replay wrong: {err1}"
    del g, tp, tp2, A, B, C


def test_gpu_graph_rejects_cpu_tasks(gctx):
    """A captured pool containing a CPU task fails capture loudly (replay
    could not reproduce host work)."""
    A = pm.TiledMatrix(gctx, 512, 512, 256, 256, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_spd_fill(tp, A, 7)
    tp.wait()
    tp2 = pm.Dtd(gctx)
    tp2.capture_begin()
    tp2.insert_py(lambda: None, flows=[(A.tile(0, 0), pm.ACCESS_IN)])
    with pytest.raises(RuntimeError, match="CPU task"):
        tp2.capture_end()
    del tp, tp2, A


def test_gpu_graph_replay_timing(gctx):
    """Replay must not be slower than re-running the pool, and reports
    per-iteration time for the launch-bound regime (many small tiles)."""
    import time
    mb = 128
    A = pm.TiledMatrix(gctx, 8 * mb, 8 * mb, mb, mb, 1, 1)
    B = pm.TiledMatrix(gctx, 8 * mb, 8 * mb, mb, mb, 1, 1)
    C = pm.TiledMatrix(gctx, 8 * mb, 8 * mb, mb, mb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_full_fill(tp, A, 3)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()
    tp2 = pm.Dtd(gctx)
    tp2.capture_begin()
    pm.insert_gemm_fp64(tp2, A, B, C)
    g = tp2.capture_end()
    g.launch(2)  # warm
    iters = 20
    t0 = time.perf_counter()
    g.launch(iters)
    graph_ms = (time.perf_counter() - t0) / iters * 1e3
    t1 = time.perf_counter()
    for _ in range(3):
        tpn = pm.Dtd(gctx)
        pm.insert_gemm_fp64(tpn, A, B, C)
        tpn.wait()
    pool_ms = (time.perf_counter() - t1) / 3 * 1e3
    print(f"graph replay {graph_ms:.3f} ms/iter vs pool re-run "
          f"{pool_ms:.3f} ms/iter ({g.n_tasks} tasks)")
    assert graph_ms < pool_ms * 1.5
    del g, tp, tp2, A, B, C


def test_jacobi_replay_example_gpu():
    """examples/jacobi_replay.py takes the hipGraph replay path on a GPU
    and converges to the same fixed point."""
    import subprocess
    r = subprocess.run([sys.executable,
                        os.path.join(REPO, "examples", "jacobi_replay.py"),
                        "4", "64", "50"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "hipGraph replay" in r.stdout, r.stdout


def test_gpu_posv_numerics(gctx):
    """insert_posv on the GPU engine (rocBLAS dtrsm solves + gemm updates)
    vs numpy.linalg.solve."""
    n, nb, nrhs = 1024, 256, 256
    A = pm.TiledMatrix(gctx, n, n, nb, nb, 1, 1, sym=True)
    B = pm.TiledMatrix(gctx, n, nrhs, nb, nb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_spd_fill(tp, A, 11)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()
    Lo = np.zeros((n, n))
    for i in range(A.mt):
        for j in range(i + 1):
            Lo[i * nb:(i + 1) * nb, j * nb:(j + 1) * nb] = A.tile_numpy(i, j)
    Lo = np.tril(Lo)
    Af = Lo + np.tril(Lo, -1).T
    Bf = full_matrix(B, nb, nb)[:, :nrhs]
    tp2 = pm.Dtd(gctx)
    pm.insert_posv(tp2, A, B)
    tp2.wait()
    X = full_matrix(B, nb, nb)[:, :nrhs]
    ref = np.linalg.solve(Af, Bf)
    err = abs(X - ref).max() / abs(ref).max()
    assert err < 1e-10, err
    del A, B, tp, tp2


def test_gpu_gesv_numerics(gctx):
    """insert_gesv_nopiv on the GPU engine vs numpy.linalg.solve."""
    n, nb, nrhs = 1024, 256, 256
    A = pm.TiledMatrix(gctx, n, n, nb, nb, 1, 1)
    B = pm.TiledMatrix(gctx, n, nrhs, nb, nb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_full_fill(tp, A, 9)
    pm.insert_full_fill(tp, B, 5)
    pm.insert_apply_scale(tp, A, 0.01, 0)
    tp.wait()
    for i in range(A.mt):
        t = A.tile_numpy(i, i)
        t += np.eye(nb) * 50.0
        A.tile_numpy_set(i, i, t)
    Af = full_matrix(A, nb, nb)
    Bf = full_matrix(B, nb, nb)[:, :nrhs]
    tp2 = pm.Dtd(gctx)
    pm.insert_gesv_nopiv(tp2, A, B)
    tp2.wait()
    ref = np.linalg.solve(Af, Bf)
    err = abs(full_matrix(B, nb, nb)[:, :nrhs] - ref).max() / abs(ref).max()
    assert err < 1e-9, err
    del A, B, tp, tp2


def test_gpu_gels_numerics(gctx):
    """insert_gels_bcgs on the GPU engine vs numpy.linalg.lstsq."""
    m, n, nb, nrhs = 2048, 1024, 256, 256
    A = pm.TiledMatrix(gctx, m, n, nb, nb, 1, 1)
    R = pm.TiledMatrix(gctx, n, n, nb, nb, 1, 1)
    B = pm.TiledMatrix(gctx, m, nrhs, nb, nb, 1, 1)
    X = pm.TiledMatrix(gctx, n, nrhs, nb, nb, 1, 1)
    tp = pm.Dtd(gctx)
    pm.insert_full_fill(tp, A, 3)
    pm.insert_full_fill(tp, B, 5)
    tp.wait()
    Af = full_matrix(A, nb, nb)[:, :n]
    Bf = full_matrix(B, nb, nb)[:, :nrhs]
    tp2 = pm.Dtd(gctx)
    pm.insert_gels_bcgs(tp2, A, R, B, X)
    tp2.wait()
    ref, *_ = np.linalg.lstsq(Af, Bf, rcond=None)
    err = abs(full_matrix(X, nb, nb)[:, :nrhs] - ref).max() / abs(ref).max()
    assert err < 1e-7, err
    del A, R, B, X, tp, tp2
