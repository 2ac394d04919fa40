"""Subprocess worker for multi-rank CPU tests (TCP comm engine).

Runs a distributed fill + Cholesky + flush_all and dumps local tiles to an
.npz for the parent to assemble and check.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import parsec_amd as pm  # noqa: E402


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    port = int(os.environ["PARSEC_TEST_PORT"])
    out = os.environ["PARSEC_TEST_OUT"]
    p, q = int(os.environ.get("GRID_P", world)), int(os.environ.get("GRID_Q", 1))
    n, nb = int(os.environ.get("MAT_N", 256)), int(os.environ.get("MAT_NB", 64))

    use_gpu = os.environ.get("PARSEC_TEST_GPU") == "1"
    pm.param_set("comm_base_port", str(port))
    ctx = pm.Context(nworkers=2, rank=rank, world=world, comm="tcp",
                     gpu=(0 if use_gpu else -2))

    app = os.environ.get("PARSEC_TEST_APP", "potrf")
    A = pm.TiledMatrix(ctx, n, n, nb, nb, p, q)
    tp = pm.Dtd(ctx)
    pm.insert_spd_fill(tp, A, 42)
    tp.wait()
    if app in ("qr", "lu", "ptglu"):
        # QR/LU need the full matrix: fill upper tiles too (spd fill
        # covers the lower triangle only; keep diagonal dominance for LU)
        import numpy as np2
        rng = np2.random.default_rng(5)
        for tm in range(A.mt):
            for tn in range(A.nt):
                if tn > tm and A.is_local(tm, tn):
                    A.tile_numpy_set(tm, tn, rng.standard_normal(
                        (A.tile_rows(tm), A.tile_cols(tn))) )
    hi = ((lambda tm: A.nt) if app in ("qr", "lu", "ptglu")
          else (lambda tm: min(tm + 1, A.nt)))
    pre = {}
    for tm in range(A.mt):
        for tn in range(hi(tm)):
            if A.is_local(tm, tn):
                pre[f"{tm}_{tn}"] = A.tile_numpy(tm, tn)

    tp2 = pm.Dtd(ctx)
    if app == "qr":
        pm.insert_geqrf(tp2, A)
    elif app == "lu":
        pm.insert_getrf_nopiv(tp2, A)
    elif app == "ptglu":
        from parsec_amd.ptg import compile_jdf
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        mod = compile_jdf(os.path.join(repo, "examples", "lu.jdf"))
        mod.build(ctx, tp2, descA=A, NT=A.mt, NB=nb)
    elif app == "ptg":
        from parsec_amd.ptg import compile_jdf
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        mod = compile_jdf(os.path.join(repo, "examples", "cholesky.jdf"))
        mod.build(ctx, tp2, descA=A, NT=A.mt, NB=nb)
    else:
        pm.insert_potrf(tp2, A)
    tp2.flush_all(A)
    tp2.wait()
    ctx.barrier()

    post = {}
    for tm in range(A.mt):
        for tn in range(hi(tm)):
            if A.is_local(tm, tn):
                post[f"{tm}_{tn}"] = A.tile_numpy(tm, tn)

    np.savez(os.path.join(out, f"rank{rank}.npz"),
             **{f"pre_{k}": v for k, v in pre.items()},
             **{f"post_{k}": v for k, v in post.items()})
    ctx.barrier()
    del ctx


if __name__ == "__main__":
    main()
