"""GPU-side debug: isolated MFMA dgemm + potf2 numerics with error pattern."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from parsec_amd import _core

def chk_gemm(m, n, k):
    rng = np.random.default_rng(0)
    A = np.asfortranarray(rng.standard_normal((m, k)))
    B = np.asfortranarray(rng.standard_normal((n, k)))
    C = np.asfortranarray(rng.standard_normal((m, n)))
    ref = C - A @ B.T
    Cf = C.copy(order="F")
    _core.dgemm_nt_hip(A.ravel(order="F"), B.ravel(order="F"), cbuf := Cf.ravel(order="F"), m, n, k)
    got = cbuf.reshape((m, n), order="F")
    err = np.abs(got - ref)
    print(f"gemm {m}x{n}x{k}: max err {err.max():.3e}")
    if err.max() > 1e-10:
        bad = np.argwhere(err > err.max() / 2)
        print("  worst rows:", sorted(set(bad[:, 0] // 16))[:10], "cols:", sorted(set(bad[:, 1] // 16))[:10])
        print("  n bad:", len(np.argwhere(err > 1e-10)), "of", m * n)
        i, j = bad[0]
        print(f"  sample [{i},{j}]: got {got[i,j]:.15f} ref {ref[i,j]:.15f}")
    return err.max()

def chk_potf2(n):
    rng = np.random.default_rng(1)
    M = rng.standard_normal((n, n))
    M = np.tril(M @ M.T + n * np.eye(n))
    buf = np.asfortranarray(M).ravel(order="F")
    _core.potf2_hip(buf, n)
    got = np.tril(buf.reshape((n, n), order="F"))
    ref = np.linalg.cholesky(np.tril(M) + np.tril(M, -1).T)
    err = np.abs(got - ref).max()
    print(f"potf2 {n}: max err {err:.3e}")
    return err

for shape in [(16,16,4), (16,16,16), (32,32,16), (128,128,16), (128,128,128), (256,128,64), (2048,2048,2048), (100,60,40)]:
    chk_gemm(*shape)
for n in [16, 64, 100, 128]:
    chk_potf2(n)

print("\n--- dgemm kernel throughput (TFLOP/s, device-resident) ---")
for (m,n,k) in [(2048,2048,2048), (4096,4096,4096), (8192,8192,2048)]:
    for impl in ["v2", "v1", "rocblas"]:
        tf = _core.bench_dgemm(m, n, k, 10, impl)
        print(f"{m}x{n}x{k} {impl}: {tf:.1f} TF/s")

print("\n--- bf16 gemm kernel throughput ---")
for (m,n,k) in [(4096,4096,4096), (8192,8192,8192)]:
    tf = _core.bench_gemm_bf16(m, n, k, 10)
    print(f"bf16 {m}x{n}x{k}: {tf:.0f} TF/s")
