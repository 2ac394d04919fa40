"""Reshape-promise engine (parsec_reshape.c analog, src/kernels_reshape.cpp):
READ flows consume lazily-materialized converted copies, shared per
{version, kind, consumer rank}."""
import os
import struct
import subprocess
import sys

import numpy as np
import pytest

import parsec_amd as pm

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)


@pytest.fixture(scope="module")
def ctx():
    c = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
    yield c
    del c


def _fill(A, tm, tn, arr):
    A.tile_numpy_set(tm, tn, arr)


def test_transpose_promise(ctx):
    nb = 24
    A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    R = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    src = np.arange(nb * nb, dtype=np.float64).reshape(nb, nb)
    _fill(A, 0, 0, src)
    tp = pm.Dtd(ctx)
    got = {}

    def body(x, out):
        v = np.frombuffer(x, dtype=np.float64).reshape(nb, nb, order="F")
        np.frombuffer(out, dtype=np.float64)[:] = v.flatten(order="F")

    tp.insert_py(body, [(A.tile(0, 0), pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE),
                        (R.tile(0, 0), pm.ACCESS_OUT)], with_data=True)
    tp.wait()
    assert np.allclose(R.tile_numpy(0, 0), src.T)


def test_tri_and_bf16_promises(ctx):
    nb = 16
    A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    src = np.random.RandomState(3).randn(nb, nb)
    _fill(A, 0, 0, src)
    tp = pm.Dtd(ctx)
    out = {}

    def grab(name):
        def body(x):
            out[name] = np.frombuffer(x, dtype=np.float64).reshape(
                nb, nb, order="F").copy()
        return body

    def grab_bf16(x):
        v = np.frombuffer(x, dtype=np.uint16).astype(np.uint32) << 16
        out["bf16"] = v.view(np.float32)[::2] if False else \
            v.copy()

    t = A.tile(0, 0)
    tp.insert_py(grab("tril"), [(t, pm.ACCESS_IN, pm.RESHAPE_TRIL)],
                 with_data=True)
    tp.insert_py(grab("triu"), [(t, pm.ACCESS_IN, pm.RESHAPE_TRIU)],
                 with_data=True)
    def bf(x):
        u = np.frombuffer(x, dtype=np.uint16).astype(np.uint32) << 16
        out["bf16"] = u.view(np.uint32).astype(np.uint32)
        out["bf16f"] = np.frombuffer(
            u.astype(np.uint32).tobytes(), dtype=np.float32).reshape(
                nb, nb, order="F").copy()
    tp.insert_py(bf, [(t, pm.ACCESS_IN, pm.RESHAPE_TO_BF16)], with_data=True)
    tp.wait()
    assert np.allclose(out["tril"], np.tril(src))
    assert np.allclose(out["triu"], np.triu(src))
    assert np.allclose(out["bf16f"], src, atol=0.02, rtol=0.02)


def test_promise_shared_across_consumers(ctx):
    """Two consumers of the same conversion share ONE conversion task;
    a version bump invalidates the promise."""
    nb = 8
    A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    _fill(A, 0, 0, np.eye(nb))
    t = A.tile(0, 0)
    base = ctx.counters()["tasks_executed_cpu"]
    tp = pm.Dtd(ctx)
    for _ in range(3):  # same version: one conversion total
        tp.insert_py(lambda x: None, [(t, pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE)],
                     with_data=True)
    tp.wait()
    n1 = ctx.counters()["tasks_executed_cpu"] - base
    assert n1 == 3 + 1, n1  # 3 consumers + 1 shared conversion
    # bump the version: the next consumer re-converts
    tp2 = pm.Dtd(ctx)
    pm.insert_apply_scale(tp2, A, 2.0, 0.0)
    tp2.insert_py(lambda x: None, [(t, pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE)],
                  with_data=True)
    tp2.wait()
    n2 = ctx.counters()["tasks_executed_cpu"] - base - n1
    assert n2 == 3, n2  # scale + consumer + fresh conversion


def test_reshape_distributed(tmp_path):
    """World-2: the consumer rank fetches the remote tile and converts it
    locally (consumer-side reshape, remote_dep_mpi.c:641-738 analog)."""
    import conftest
    port = str(conftest.port_base(25))
    code = """
import os, sys
sys.path.insert(0, %r)
import numpy as np
import parsec_amd as pm
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
nb = 12
A = pm.TiledMatrix(ctx, 2 * nb, nb, nb, nb, 2, 1)   # tile(0,0) on rank 0
src = np.arange(nb * nb, dtype=np.float64).reshape(nb, nb)
if A.is_local(0, 0):
    A.tile_numpy_set(0, 0, src)
tp = pm.Dtd(ctx)
seen = {}
def body(x):
    seen["T"] = np.frombuffer(x, dtype=np.float64).reshape(
        nb, nb, order="F").copy()
# consumer on rank 1 with a transpose promise on rank 0's tile
tp.insert_py(body, [(A.tile(0, 0), pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE)],
             rank=1, with_data=True)
tp.wait()
if rank == 1:
    assert np.allclose(seen["T"], src.T), seen["T"]
    print("RESHAPE_DIST_OK")
ctx.barrier()
del A, ctx
""" % (REPO,)
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        outs.append(o)
        assert pr.returncode == 0, o.decode()
    assert b"RESHAPE_DIST_OK" in outs[1]


@pytest.mark.gpu
def test_reshape_gpu_numerics():
    """GPU conversion kernels vs numpy."""
    c = pm.Context(nworkers=2, rank=0, world=1)
    assert c.has_gpu
    nb = 96
    A = pm.TiledMatrix(c, nb, nb, nb, nb, 1, 1)
    R = pm.TiledMatrix(c, nb, nb, nb, nb, 1, 1)
    src = np.random.RandomState(5).randn(nb, nb)
    A.tile_numpy_set(0, 0, src)
    tp = pm.Dtd(c)
    # run a GPU task between fill and consume so the tile is device-resident
    pm.insert_apply_scale(tp, A, 1.0, 0.0)

    def body(x, out):
        np.frombuffer(out, dtype=np.float64)[:] = np.frombuffer(
            x, dtype=np.float64)

    tp.insert_py(body, [(A.tile(0, 0), pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE),
                        (R.tile(0, 0), pm.ACCESS_OUT)], with_data=True)
    tp.wait()
    assert c.counters()["tasks_executed_gpu"] > 0
    assert np.allclose(R.tile_numpy(0, 0), src.T)
    del A, R, c


def test_reshape_after_rename(ctx):
    """Reshape promises attach to the CURRENT copy: after an OUTPUT-only
    rewrite forces a rename, the next promise converts the new version."""
    nb = 8
    A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    _fill(A, 0, 0, np.ones((nb, nb)))
    t = A.tile(0, 0)
    got = []
    tp = pm.Dtd(ctx)
    for k in range(4):
        # slow reader keeps the old copy live -> the OUT rewrite renames
        def reader(x, k=k):
            import time
            time.sleep(0.01)
            got.append(np.frombuffer(x, dtype=np.float64)[0])
        tp.insert_py(reader, [(t, pm.ACCESS_IN, pm.RESHAPE_TRIL)],
                     with_data=True)
        def writer(buf, k=k):
            np.frombuffer(buf, dtype=np.float64)[:] = float(k + 2)
        tp.insert_py(writer, [(t, pm.ACCESS_OUT)], with_data=True)
    tp.wait()
    # renaming makes readers of successive versions CONCURRENT, so the
    # completion order is free — each must still see its own version
    assert sorted(got) == [1.0, 2.0, 3.0, 4.0], got
    assert ctx.counters()["renames"] > 0


def test_wait_dynamic_repeated(ctx):
    """wait_dynamic is reusable (handler teardown between calls)."""
    tp = pm.Dtd(ctx)
    acc = []
    for r in range(3):
        for i in range(10):
            tp.insert_py(lambda i=i: acc.append(i))
        tp.wait_dynamic()
    assert len(acc) == 30
