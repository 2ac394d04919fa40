"""bf16 MFMA tile-GEMM DAG (BASELINE config 5) numerics."""
import numpy as np
import pytest

import parsec_amd as pm
from parsec_amd import _core


def to_bf16_bits(x):
    u = x.astype(np.float32).view(np.uint32)
    r = ((u + 0x7FFF + ((u >> 16) & 1)) >> 16).astype(np.uint16)
    return r


def bf16_to_f32(b):
    return (b.astype(np.uint32) << 16).view(np.float32)


def test_bf16_dag_cpu(ctx):
    K, M, N, kb, mb = 128, 128, 128, 64, 64
    At = pm.TiledMatrix(ctx, K, M, kb, mb, 1, 1, elem_size=2)
    B = pm.TiledMatrix(ctx, K, N, kb, mb, 1, 1, elem_size=2)
    C = pm.TiledMatrix(ctx, M, N, mb, mb, 1, 1, elem_size=4)
    tp = pm.Dtd(ctx)
    pm.insert_fill_bf16(tp, At, 1)
    pm.insert_fill_bf16(tp, B, 2)
    pm.insert_gemm_bf16(tp, At, B, C)
    tp.wait()
    # assemble in fp32 and compare
    def asm_bf(T, rows, cols, rb, cb):
        M_ = np.zeros((rows, cols), dtype=np.float32)
        for tm in range(T.mt):
            for tn in range(T.nt):
                raw = np.frombuffer(T.tile_bytes(tm, tn), dtype=np.uint16)
                tile = bf16_to_f32(raw).reshape((cb, rb)).T  # col-major
                M_[tm * rb:(tm + 1) * rb, tn * cb:(tn + 1) * cb] = tile
        return M_

    Am = asm_bf(At, K, M, kb, mb)
    Bm = asm_bf(B, K, N, kb, mb)
    ref = Am.T.astype(np.float64) @ Bm.astype(np.float64)
    got = np.zeros((M, N), dtype=np.float32)
    for tm in range(C.mt):
        for tn in range(C.nt):
            raw = np.frombuffer(C.tile_bytes(tm, tn), dtype=np.float32)
            got[tm * mb:(tm + 1) * mb, tn * mb:(tn + 1) * mb] = \
                raw.reshape((mb, mb)).T
    rel = np.abs(got - ref).max() / max(1e-6, np.abs(ref).max())
    assert rel < 1e-4, f"bf16 DAG rel err {rel}"


@pytest.mark.gpu
def test_bf16_kernel_vs_numpy():
    rng = np.random.default_rng(0)
    m, n, k = 256, 192, 320
    A = to_bf16_bits(rng.standard_normal((m, k)))   # row i, col kk
    B = to_bf16_bits(rng.standard_normal((n, k)))
    # kernel wants col-major k x m (= row-major m x k buffer) — A as laid
    # out row-major IS col-major (k fastest per m): pass flattened
    C = np.zeros((n, m), dtype=np.float32)  # col-major m x n
    _core.gemm_bf16_hip(A.ravel(), B.ravel(), C.ravel(), m, n, k)
    got = C.T  # back to m x n
    ref = bf16_to_f32(A).astype(np.float64) @ bf16_to_f32(B).astype(np.float64).T
    rel = np.abs(got - ref).max() / np.abs(ref).max()
    assert rel < 1e-4, f"bf16 kernel rel err {rel}"


@pytest.mark.gpu
def test_bf16_dag_gpu():
    ctx = pm.Context(nworkers=2, rank=0, world=1)
    K, M, N, kb, mb = 1024, 512, 512, 256, 256
    At = pm.TiledMatrix(ctx, K, M, kb, mb, 1, 1, elem_size=2)
    B = pm.TiledMatrix(ctx, K, N, kb, mb, 1, 1, elem_size=2)
    C = pm.TiledMatrix(ctx, M, N, mb, mb, 1, 1, elem_size=4)
    tp = pm.Dtd(ctx)
    pm.insert_fill_bf16(tp, At, 1)
    pm.insert_fill_bf16(tp, B, 2)
    pm.insert_gemm_bf16(tp, At, B, C)
    tp.wait()
    c00 = np.frombuffer(C.tile_bytes(0, 0), dtype=np.float32)
    assert np.isfinite(c00).all() and np.abs(c00).max() > 0
    del At, B, C, ctx


def test_bf16_dag_distributed(tmp_path):
    """Config-5 DAG across 2 ranks (TCP engine): bf16 operand tiles and
    fp32 C tiles flow between ranks through the SPMD protocol."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = """
import os, sys
sys.path.insert(0, {repo!r})
import numpy as np
import parsec_amd as pm
rank = int(os.environ["RANK"]); world = 2
pm.param_set("comm_base_port", os.environ["PARSEC_TEST_PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=world, comm="tcp", gpu=-2)
K, M, N, kb, mb = 128, 128, 128, 64, 64
At = pm.TiledMatrix(ctx, K, M, kb, mb, 2, 1, elem_size=2)
B = pm.TiledMatrix(ctx, K, N, kb, mb, 2, 1, elem_size=2)
C = pm.TiledMatrix(ctx, M, N, mb, mb, 2, 1, elem_size=4)
tp = pm.Dtd(ctx)
pm.insert_fill_bf16(tp, At, 1)
pm.insert_fill_bf16(tp, B, 2)
pm.insert_gemm_bf16(tp, At, B, C)
tp.flush_all(C)
tp.wait()
ctx.barrier()
out = {{}}
def bf2f(b):
    return (np.frombuffer(b, dtype=np.uint16).astype(np.uint32) << 16).view(np.float32)
for T, nm in [(At, "A"), (B, "B")]:
    for tm in range(T.mt):
        for tn in range(T.nt):
            if T.is_local(tm, tn):
                out[f"{{nm}}_{{tm}}_{{tn}}"] = bf2f(T.tile_bytes(tm, tn))
for tm in range(C.mt):
    for tn in range(C.nt):
        if C.is_local(tm, tn):
            out[f"C_{{tm}}_{{tn}}"] = np.frombuffer(C.tile_bytes(tm, tn), dtype=np.float32)
np.savez(os.path.join({out!r}, f"bf16_rank{{rank}}.npz"), **out)
ctx.barrier()
del At, B, C, ctx
""".format(repo=repo, out=str(tmp_path))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env["RANK"] = str(r)
        env["WORLD_SIZE"] = "2"
        env["PARSEC_TEST_PORT"] = str(__import__("conftest").port_base(7))
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        assert pr.returncode == 0, o.decode()
    K, M, N, kb, mb = 128, 128, 128, 64, 64
    Am = np.zeros((K, M), dtype=np.float64)
    Bm = np.zeros((K, N), dtype=np.float64)
    Cm = np.zeros((M, N), dtype=np.float64)
    for r in range(2):
        z = np.load(os.path.join(tmp_path, f"bf16_rank{r}.npz"))
        for key in z.files:
            nm, tm, tn = key.split("_")
            tm, tn = int(tm), int(tn)
            dst, rb, cb = {"A": (Am, kb, mb), "B": (Bm, kb, mb),
                           "C": (Cm, mb, mb)}[nm]
            dst[tm * rb:(tm + 1) * rb, tn * cb:(tn + 1) * cb] = \
                z[key].reshape((cb, rb)).T
    ref = Am.T @ Bm
    rel = np.abs(Cm - ref).max() / max(1e-6, np.abs(ref).max())
    assert rel < 1e-4, f"distributed bf16 rel err {rel}"
