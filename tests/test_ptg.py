"""PTG DSL tests: parsec_ptgpp compiles .jdf files and the generated
taskpools execute with correct dataflow semantics.

Mirrors the reference's tests/dsl/ptg tree (chains, CTL ordering, guards/
ternaries, the must-fail-to-compile compiler tests).
"""
import os
import subprocess
import sys
import struct

import numpy as np
import pytest

import parsec_amd as pm
from parsec_amd.ptg import JdfError, compile_jdf

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EX = os.path.join(REPO, "examples")


def test_chain_sequential(ctx):
    mod = compile_jdf(os.path.join(EX, "Ex02_Chain.jdf"))
    A = pm.TiledMatrix(ctx, 1, 1, 1, 1, 1, 1)  # one 8-byte tile
    A.tile_bytes_set(0, 0, struct.pack("<q", 0))
    tp = pm.Dtd(ctx, "chain")
    mod.build(ctx, tp, mydata=A, NT=10)
    tp.wait()
    (v,) = struct.unpack("<q", A.tile_bytes(0, 0))
    assert v == 2**10 - 1  # v = 2v+1 ten times, sequentially


def test_ctl_chain_orders(ctx):
    mod = compile_jdf(os.path.join(EX, "ctl_chain.jdf"))
    NT = 8
    A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 1, 1)
    tp = pm.Dtd(ctx, "ctl")
    mod.build(ctx, tp, mydata=A, NT=NT)
    tp.wait()
    seqs = [struct.unpack("<q", A.tile_bytes(k, 0))[0] for k in range(NT)]
    # pure CTL chain: tasks have disjoint tiles, yet must run in order
    assert sorted(seqs) == seqs, f"CTL ordering violated: {seqs}"


def test_ptg_cholesky_vs_numpy(ctx):
    n, nb = 256, 64
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    tp0 = pm.Dtd(ctx)
    pm.insert_spd_fill(tp0, A, 42)
    tp0.wait()
    M = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(tm + 1):
            M[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = A.tile_numpy(tm, tn)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)

    mod = compile_jdf(os.path.join(EX, "cholesky.jdf"))
    tp = pm.Dtd(ctx, "ptg_potrf")
    mod.build(ctx, tp, descA=A, NT=A.mt, NB=nb)
    tp.wait()
    L = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(tm + 1):
            L[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = A.tile_numpy(tm, tn)
    err = np.abs(np.tril(L) - L0).max()
    assert err < 1e-10, f"PTG cholesky max err {err}"


def _write_tmp_jdf(tmp_path, text):
    p = tmp_path / "bad.jdf"
    p.write_text(text)
    return str(p)


def test_must_fail_cuda_body(ctx, tmp_path):
    bad = """
mydata [ type="parsec_data_collection_t*" ]
T(k)
k = 0 .. 3
: mydata( k )
RW A <- mydata( k )
BODY [type=CUDA]
{ }
END
"""
    with pytest.raises(JdfError, match="MI355X-native"):
        compile_jdf(_write_tmp_jdf(tmp_path, bad))


def test_must_fail_unknown_class(ctx, tmp_path):
    bad = """
mydata [ type="parsec_data_collection_t*" ]
T(k)
k = 0 .. 3
: mydata( k )
RW A <- A NoSuchTask( k )
BODY
{ }
END
"""
    with pytest.raises(JdfError, match="unknown task class"):
        compile_jdf(_write_tmp_jdf(tmp_path, bad))


def test_must_fail_syntax(ctx, tmp_path):
    with pytest.raises(JdfError):
        compile_jdf(_write_tmp_jdf(tmp_path, "T(k)\nk = 0 ..\n???"))


@pytest.mark.gpu
def test_ptg_hip_body(ctx):
    mod = compile_jdf(os.path.join(EX, "scale_hip.jdf"))
    NT, nelem = 4, 1024
    A = pm.TiledMatrix(ctx, NT * nelem, 1, nelem, 1, 1, 1)
    for k in range(NT):
        A.tile_numpy_set(k, 0, np.full((nelem, 1), float(k)))
    tp = pm.Dtd(ctx, "scale")
    mod.build(ctx, tp, descA=A, NT=NT, NELEM=nelem)
    tp.wait()
    for k in range(NT):
        got = A.tile_numpy(k, 0)
        assert np.allclose(got, k * 2.0 + (k + 1)), f"tile {k} wrong"


@pytest.mark.gpu
def test_ptg_hip_cholesky():
    """PTG Cholesky with rocBLAS/rocSOLVER HIP bodies (stress.jdf analog)."""
    n, nb = 1024, 256
    ctx2 = pm.Context(nworkers=2, rank=0, world=1)
    A = pm.TiledMatrix(ctx2, n, n, nb, nb, 1, 1)
    tp0 = pm.Dtd(ctx2)
    pm.insert_spd_fill(tp0, A, 42)
    tp0.wait()
    M = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(tm + 1):
            M[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = A.tile_numpy(tm, tn)
    M = np.tril(M) + np.tril(M, -1).T
    L0 = np.linalg.cholesky(M)
    mod = compile_jdf(os.path.join(EX, "cholesky_hip.jdf"))
    tp = pm.Dtd(ctx2, "ptg_hip_potrf")
    mod.build(ctx2, tp, descA=A, NT=A.mt, NB=nb)
    tp.wait()
    L = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(tm + 1):
            L[tm * nb:(tm + 1) * nb, tn * nb:(tn + 1) * nb] = A.tile_numpy(tm, tn)
    err = np.abs(np.tril(L) - L0).max()
    assert err < 1e-8, f"PTG HIP cholesky max err {err}"
    del A, ctx2


def test_ptg_distributed(tmp_path):
    """Multi-rank PTG: the generated taskpool's deterministic enumeration +
    topo order is identical on every rank, so the SPMD dataflow protocol
    carries JDF programs across ranks unchanged (world 2, TCP engine)."""
    import subprocess
    import sys as _sys
    # warm the compile cache to keep the subprocesses fast
    compile_jdf(os.path.join(EX, "cholesky.jdf"))
    worker = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "_dist_worker.py")
    world, n, nb = 2, 256, 64
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE=str(world),
                   PARSEC_TEST_PORT=str(__import__("conftest").port_base(5)), PARSEC_TEST_OUT=str(tmp_path),
                   GRID_P="2", GRID_Q="1", MAT_N=str(n), MAT_NB=str(nb),
                   PARSEC_TEST_APP="ptg")
        procs.append(subprocess.Popen([_sys.executable, worker], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        out, _ = pr.communicate(timeout=300)
        assert pr.returncode == 0, f"worker failed:\n{out.decode()}"
    pre = np.zeros((n, n))
    post = np.zeros((n, n))
    for r in range(world):
        z = np.load(os.path.join(tmp_path, f"rank{r}.npz"))
        for key in z.files:
            kind, tm, tn = key.split("_")[0], *key.split("_")[1:]
            v = z[key]
            dst = pre if kind == "pre" else post
            tm, tn = int(tm), int(tn)
            dst[tm * nb:tm * nb + v.shape[0], tn * nb:tn * nb + v.shape[1]] = v
    L0 = np.linalg.cholesky(np.tril(pre) + np.tril(pre, -1).T)
    err = np.abs(np.tril(post) - L0).max()
    assert err < 1e-10, f"distributed PTG max err {err}"


def test_inline_c_and_hidden_globals(ctx):
    import struct
    mod = compile_jdf(os.path.join(EX, "inline_c.jdf"))
    NT = 6
    A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 1, 1)
    tp = pm.Dtd(ctx, "probe")
    mod.build(ctx, tp, mydata=A, NT=NT)
    tp.wait()
    for k in range(NT):
        (v,) = struct.unpack("<q", A.tile_bytes(k, 0))
        assert v == k * 2 + NT // 2, (k, v)


def test_new_tiles(ctx):
    import struct
    mod = compile_jdf(os.path.join(EX, "new_tile.jdf"))
    NT = 5
    A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 1, 1)
    tp = pm.Dtd(ctx, "newt")
    mod.build(ctx, tp, mydata=A, NT=NT)
    tp.wait()
    for k in range(NT):
        (v,) = struct.unpack("<q", A.tile_bytes(k, 0))
        assert v == sum(k * 100 + i for i in range(8)), (k, v)


def test_ctl_cross_rank(tmp_path):
    """CTL flows serialize across ranks (tokens ride the dataflow)."""
    import subprocess
    import sys as _sys
    jdf = """
mydata  [ type="parsec_data_collection_t*" ]
NT      [ type="int" ]

Step(k)

k = 0 .. NT-1

: mydata( k )

WRITE A -> mydata( k )
CTL  X <- (k > 0) ? X Step( k-1 )
       -> (k < NT-1) ? X Step( k+1 )

BODY
{
    long* v = (long*)A;
    v[0] = k;
}
END
"""
    jp = tmp_path / "xrank_ctl.jdf"
    jp.write_text(jdf)
    compile_jdf(str(jp))  # warm cache
    code = f"""
import os, sys, struct
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
NT = 6
A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 2, 1)  # tiles alternate ranks
mod = compile_jdf({str(jp)!r})
tp = pm.Dtd(ctx, "xctl")
mod.build(ctx, tp, mydata=A, NT=NT)
tp.wait()
ctx.barrier()
for k in range(NT):
    if A.is_local(k, 0):
        (v,) = struct.unpack("<q", A.tile_bytes(k, 0))
        assert v == k, (k, v)
print("XRANK_CTL_OK", rank)
ctx.barrier()
del A, ctx
"""
    import conftest
    port = str(conftest.port_base(9))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([_sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        assert pr.returncode == 0 and b"XRANK_CTL_OK" in o, o.decode()


def test_broadcast_world2(tmp_path):
    """One-to-many broadcast: sent-mask dedup => one send per dest rank."""
    import subprocess
    import sys as _sys
    code = f"""
import os, sys, struct
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
NT = 8
A = pm.TiledMatrix(ctx, NT, 4, 1, 4, 2, 1)
mod = compile_jdf(os.path.join({EX!r}, "broadcast.jdf"))
tp = pm.Dtd(ctx, "bcast")
mod.build(ctx, tp, mydata=A, NT=NT)
tp.wait()
ctx.barrier()
for k in range(NT):
    if A.is_local(k, 0):
        vals = struct.unpack("<4d", A.tile_bytes(k, 0))
        assert vals == tuple(10.0 + i + k for i in range(4)), (k, vals)
# root sent the tile to rank 1 exactly once (plus readers on rank 0 free)
c = ctx.counters()
if rank == 0:
    assert c["comm_msgs"] <= 2, c  # one payload (+barrier accounting)
print("BCAST_OK", rank)
ctx.barrier()
del A, ctx
"""
    import conftest
    port = str(conftest.port_base(15))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([_sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        assert pr.returncode == 0 and b"BCAST_OK" in o, o.decode()


def test_ctl_gather(ctx):
    """Ranged CTL fan-in (ctlgat.jdf): Gather runs after ALL producers."""
    import struct
    mod = compile_jdf(os.path.join(EX, "ctl_gather.jdf"))
    NT = 12
    A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 1, 1)
    tp = pm.Dtd(ctx, "gat")
    mod.build(ctx, tp, mydata=A, NT=NT)
    tp.wait()
    (g,) = struct.unpack("<q", A.tile_bytes(0, 0))
    assert g == 1000 + NT, g  # all NT producers retired before the gather


def test_ctl_gather_world2(tmp_path):
    """Cross-rank CTL gather: producers on both ranks precede the gather."""
    import subprocess
    import sys as _sys
    code = f"""
import os, sys, struct
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
NT = 8
A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 2, 1)
mod = compile_jdf(os.path.join({EX!r}, "ctl_gather.jdf"))
tp = pm.Dtd(ctx, "gat2")
mod.build(ctx, tp, mydata=A, NT=NT)
tp.wait()
ctx.barrier()
if A.is_local(0, 0):
    n_local = sum(1 for k in range(NT) if A.is_local(k, 0))
    (g,) = struct.unpack("<q", A.tile_bytes(0, 0))
    assert g == 1000 + n_local, (g, n_local)
print("GATHER_OK", rank)
ctx.barrier()
del A, ctx
"""
    import conftest
    port = str(conftest.port_base(17))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([_sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        assert pr.returncode == 0 and b"GATHER_OK" in o, o.decode()


def test_ptg_lu_vs_numpy(ctx):
    """LU nopiv JDF (examples/lu.jdf): L*U reconstructs A."""
    import numpy as np
    mod = compile_jdf(os.path.join(EX, "lu.jdf"))
    n, nb = 256, 64
    A = pm.TiledMatrix(ctx, n, n, nb, nb, 1, 1)
    tp = pm.Dtd(ctx, "ptglu_fill")
    pm.insert_full_fill(tp, A, 7)
    tp.wait()
    A0 = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(A.nt):
            A0[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
    tp2 = pm.Dtd(ctx, "ptglu")
    mod.build(ctx, tp2, descA=A, NT=A.mt, NB=nb)
    tp2.wait()
    F = np.zeros((n, n))
    for tm in range(A.mt):
        for tn in range(A.nt):
            F[tm*nb:(tm+1)*nb, tn*nb:(tn+1)*nb] = A.tile_numpy(tm, tn)
    L = np.tril(F, -1) + np.eye(n)
    U = np.triu(F)
    err = np.abs(L @ U - A0).max() / np.abs(A0).max()
    assert err < 1e-11, err


def test_branching_choice(ctx):
    """Guard + else-ternary dataflow routing (branching/choice analog)."""
    import struct
    mod = compile_jdf(os.path.join(EX, "branching.jdf"))
    NT = 9
    A = pm.TiledMatrix(ctx, NT, 1, 1, 1, 1, 1)
    for k in range(NT):
        A.tile_bytes_set(k, 0, struct.pack("<d", float(k)))
    tp = pm.Dtd(ctx, "branch")
    mod.build(ctx, tp, mydata=A, NT=NT)
    tp.wait()
    for k in range(NT):
        (v,) = struct.unpack("<d", A.tile_bytes(k, 0))
        want = (k + 1) * 10.0 if k % 2 == 1 else -(k + 1.0)
        assert v == want, (k, v, want)


def test_ptg_lu_distributed(tmp_path):
    """World-2 LU through the JDF path (PTG multi-rank, full-matrix app)."""
    import subprocess
    import sys as _sys
    import numpy as np
    import conftest
    worker = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "_dist_worker.py")
    world, n, nb = 2, 256, 64
    port = str(conftest.port_base(27))
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE=str(world),
                   PARSEC_TEST_PORT=port, PARSEC_TEST_OUT=str(tmp_path),
                   GRID_P="2", GRID_Q="1", MAT_N=str(n), MAT_NB=str(nb),
                   PARSEC_TEST_APP="ptglu")
        procs.append(subprocess.Popen([_sys.executable, worker], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        out, _ = pr.communicate(timeout=300)
        assert pr.returncode == 0, f"worker failed:\n{out.decode()}"
    pre = np.zeros((n, n))
    post = np.zeros((n, n))
    for r in range(world):
        z = np.load(os.path.join(tmp_path, f"rank{r}.npz"))
        for key in z.files:
            kind, tm, tn = key.split("_")[0], *key.split("_")[1:]
            tm, tn = int(tm), int(tn)
            v = z[key]
            dst = pre if kind == "pre" else post
            dst[tm * nb:tm * nb + v.shape[0], tn * nb:tn * nb + v.shape[1]] = v
    L = np.tril(post, -1) + np.eye(n)
    U = np.triu(post)
    err = np.abs(L @ U - pre).max() / np.abs(pre).max()
    assert err < 1e-11, f"PTG distributed LU rel err {err}"


# ------------------------- ptgpp must-fail battery -------------------------
# The reference keeps a compiler-test directory of JDFs that must FAIL to
# compile (tests/dsl/ptg/ptgpp/, run via NODEFAULTBUILD); these mirror that
# contract against parsec_amd's compiler diagnostics.
_GOOD_HEADER = 'A  [ type="parsec_data_collection_t*" ]\nNT [ type="int" ]\n'


def _mk(tmp_path, body):
    p = tmp_path / "mf.jdf"
    p.write_text(_GOOD_HEADER + body)
    return str(p)


MUST_FAIL = [
    # (name, jdf body, expected message fragment)
    ("too_many_params",
     "T(a,b,c,d,e,f,g,h,i)\n" + "".join(f"{v} = 0 .. 1\n" for v in
                                        "abcdefghi") +
     ": A(a, 0)\nRW X <- A(a, 0) -> A(a, 0)\nBODY\n{}\nEND\n",
     "too many parameters"),
    ("param_without_range",
     "T(k, j)\nk = 0 .. NT-1\n: A(k, 0)\n"
     "RW X <- A(k, 0) -> A(k, 0)\nBODY\n{}\nEND\n",
     "no range"),
    ("duplicate_class",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n"
     "BODY\n{}\nEND\n"
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n"
     "BODY\n{}\nEND\n",
     "duplicate task class"),
    ("duplicate_flow",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n"
     "RW X <- A(k, 1) -> A(k, 1)\nBODY\n{}\nEND\n",
     "duplicate flow"),
    ("dep_arity_mismatch",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\n"
     "RW X <- (k > 0) ? X T(k-1, 0) : A(k, 0) -> A(k, 0)\n"
     "BODY\n{}\nEND\n",
     "parameter"),
    ("dep_unknown_flow",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\n"
     "RW X <- (k > 0) ? Y T(k-1) : A(k, 0) -> A(k, 0)\nBODY\n{}\nEND\n",
     "no flow"),
    ("dep_unknown_class",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\n"
     "RW X <- (k > 0) ? X U(k-1) : A(k, 0) -> A(k, 0)\nBODY\n{}\nEND\n",
     "unknown task class"),
    ("body_without_end",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\nBODY\n{}\n",
     "without END"),
    ("class_without_body",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n",
     "no BODY"),
    ("missing_partition",
     "T(k)\nk = 0 .. 1\nRW X <- A(k, 0) -> A(k, 0)\nBODY\n{}\nEND\n",
     "partition"),
    ("new_without_size",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nWRITE X <- NEW -> A(k, 0)\n"
     "BODY\n{}\nEND\n",
     "size"),
    ("ranged_in_on_data",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\n"
     "RW X <- X T(0 .. 1) -> A(k, 0)\nBODY\n{}\nEND\n",
     "ranged"),
    ("junk_before_arrows",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X junk <- A(k, 0)\nBODY\n{}\nEND\n",
     None),
    ("unbalanced_parens",
     "T(k)\nk = 0 .. (NT-1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n"
     "BODY\n{}\nEND\n",
     "unbalanced"),
    ("cuda_body_rejected",
     "T(k)\nk = 0 .. 1\n: A(k, 0)\nRW X <- A(k, 0) -> A(k, 0)\n"
     "BODY [type=CUDA]\n{}\nEND\n",
     "MI355X-native"),
    ("unknown_collection",
     "T(k)\nk = 0 .. 1\n: B(k, 0)\nRW X <- B(k, 0) -> B(k, 0)\n"
     "BODY\n{}\nEND\n",
     "unknown data collection"),
]


@pytest.mark.parametrize("name,body,frag",
                         MUST_FAIL, ids=[m[0] for m in MUST_FAIL])
def test_ptgpp_must_fail(tmp_path, name, body, frag):
    from parsec_amd.ptg import compile_jdf, JdfError
    path = _mk(tmp_path, body)
    with pytest.raises(JdfError) as ei:
        compile_jdf(path)
    if frag:
        assert frag.lower() in str(ei.value).lower(), str(ei.value)


def test_multiline_expressions(ctx, tmp_path):
    """Ranges, guards and dependency terms spanning physical lines (the
    reference grammar is token-based; the line joiner restores that)."""
    jdf = """
A  [ type="parsec_data_collection_t*" ]
NT [ type="int" ]

Step(k)

k = 0 .. (NT
          - 1)

: A( k,
     0 )

RW X <- (k > 0)
         ? X Step(
               k - 1)
         : A(k, 0)
     -> A(k,
          0)

BODY
{
    ((long*)X)[0] += 1;
}
END
"""
    p = tmp_path / "multiline.jdf"
    p.write_text(jdf)
    from parsec_amd.ptg import compile_jdf
    import struct
    mod = compile_jdf(str(p))
    A = pm.TiledMatrix(ctx, 4, 1, 1, 1, 1, 1)
    A.tile_bytes_set(0, 0, struct.pack("<q", 0))
    tp = pm.Dtd(ctx, "ml")
    mod.build(ctx, tp, A=A, NT=4)
    tp.wait()
    (v,) = struct.unpack("<q", A.tile_bytes(0, 0))
    assert v == 4, v


def test_ptg_compact_iteration_vs_materialized():
    """Compact (never-materialized) iteration — jdf2c.c:3047+ analog:
    seeds from an O(1)-memory scan, the rest instantiated when their last
    predecessor completes via the OUT-arrow duals. Must equal the
    materialized build bit-for-bit."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import numpy as np
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
import os
import subprocess
import sys
ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 640, 640, 64, 64, 1, 1)
B = pm.TiledMatrix(ctx, 640, 640, 64, 64, 1, 1)
tp0 = pm.Dtd(ctx)
pm.insert_spd_fill(tp0, A, 42); pm.insert_spd_fill(tp0, B, 42); tp0.wait()
mod = compile_jdf(os.path.join({REPO!r}, "examples", "cholesky.jdf"))
tp1 = pm.Dtd(ctx); mod.build(ctx, tp1, descA=A, NT=A.mt, NB=A.nb); tp1.wait()
tp2 = pm.Dtd(ctx)
mod.build(ctx, tp2, compact=True, descA=B, NT=B.mt, NB=B.nb); tp2.wait()
for i in range(A.mt):
    for j in range(i + 1):
        assert np.allclose(A.tile_numpy(i, j), B.tile_numpy(i, j)), (i, j)
print("COMPACT_OK")
del A, B, tp0, tp1, tp2, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0 and "COMPACT_OK" in r.stdout, \
        r.stdout + r.stderr
    assert "WARNING" not in r.stderr, r.stderr


def test_ptg_compact_rejects_multiprocess():
    """Compact iteration is world-1 only (distributed PTG keeps the
    materialized deterministic insertion order)."""
    from conftest import port_base
    code = f"""
import os, sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 2, 1)
tp = pm.Dtd(ctx)
mod = compile_jdf(os.path.join({REPO!r}, "examples", "cholesky.jdf"))
mod.build(ctx, tp, compact=True, descA=A, NT=A.mt, NB=A.nb)
"""
    import os as _os
    port = str(port_base(31))
    procs = []
    for rk in range(2):
        env = dict(_os.environ)
        env.update(RANK=str(rk), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=120)[0] for p in procs]
    assert any(p.returncode != 0 for p in procs)
    assert any(b"single-process only" in o for o in outs), outs


def test_ptg_compact_lu():
    """Compact iteration on a second app (lu.jdf — its arrows already
    carry the full duals): equals the materialized build."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import numpy as np
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
import os
ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 384, 384, 64, 64, 1, 1)
B = pm.TiledMatrix(ctx, 384, 384, 64, 64, 1, 1)
tp0 = pm.Dtd(ctx)
pm.insert_full_fill(tp0, A, 7); pm.insert_full_fill(tp0, B, 7)
pm.insert_apply_scale(tp0, A, 0.01, 0); pm.insert_apply_scale(tp0, B, 0.01, 0)
tp0.wait()
for i in range(A.mt):
    for M in (A, B):
        t = M.tile_numpy(i, i); t += np.eye(64) * 100.0
        M.tile_numpy_set(i, i, t)
mod = compile_jdf(os.path.join({REPO!r}, "examples", "lu.jdf"))
tp1 = pm.Dtd(ctx); mod.build(ctx, tp1, descA=A, NT=A.mt, NB=A.nb); tp1.wait()
tp2 = pm.Dtd(ctx)
mod.build(ctx, tp2, compact=True, descA=B, NT=B.mt, NB=B.nb); tp2.wait()
for i in range(A.mt):
    for j in range(A.nt):
        assert np.allclose(A.tile_numpy(i, j), B.tile_numpy(i, j)), (i, j)
print("LU_COMPACT_OK")
del A, B, tp0, tp1, tp2, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0 and "LU_COMPACT_OK" in r.stdout, \
        r.stdout + r.stderr
    assert "WARNING" not in r.stderr, r.stderr


def test_ptg_compact_warns_on_missing_duals(tmp_path):
    """A JDF whose OUT arrows don't cover the IN arrows (missing duals)
    must complete what it can and WARN loudly about never-activated
    instances instead of silently dropping them."""
    jdf = """
A  [ type="parsec_data_collection_t*" ]
NT [ type="int" ]

Step(k)

k = 0 .. NT-1

: A( 0, 0 )

RW X <- (k == 0) ? A( 0, 0 ) : X Step(k - 1)

BODY
{
}
END
"""
    p = tmp_path / "nodual.jdf"
    p.write_text(jdf)
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import struct
import parsec_amd as pm
from parsec_amd.ptg import compile_jdf
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 1, 1, 1, 1, 1, 1)
A.tile_bytes_set(0, 0, struct.pack("<q", 0))
mod = compile_jdf({str(p)!r})
tp = pm.Dtd(ctx)
mod.build(ctx, tp, compact=True, A=A, NT=8)
tp.wait()
print("DRAINED")
del A, tp, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0 and "DRAINED" in r.stdout, r.stdout + r.stderr
    assert "not duals" in r.stderr, r.stderr
