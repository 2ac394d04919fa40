"""Runtime-semantics tests: DTD ordering (RAW/WAR/WAW), priorities, window.

Mirrors the reference's tests/dsl/dtd/* programs (dtd_test_war.c etc.,
SURVEY.md §4) using Python task bodies on the CPU path.
"""
import os

import pytest
import threading

import parsec_amd as pm

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_context_props(ctx):
    assert ctx.rank == 0
    assert ctx.world == 1
    assert ctx.nworkers == 2


def test_param_roundtrip(ctx):
    pm.param_set("test_param_x", "17")
    pm.Dtd(ctx)  # registers the dtd params
    assert "dtd_window_size" in pm.param_dump()


def _matrix(ctx, nt=2):
    return pm.TiledMatrix(ctx, 64 * nt, 64 * nt, 64, 64, 1, 1)


def test_raw_chain_order(ctx):
    """N writers to one tile must run in insertion order (RAW/WAW chain)."""
    A = _matrix(ctx)
    t = A.tile(0, 0)
    tp = pm.Dtd(ctx)
    log = []
    lock = threading.Lock()

    def body(i):
        with lock:
            log.append(i)

    for i in range(50):
        tp.insert_py((lambda i=i: body(i)), flows=[(t, pm.ACCESS_INOUT)])
    tp.wait()
    assert log == list(range(50))


def test_war_readers_before_writer(ctx):
    """Readers of version v all run before the next writer (WAR edges)."""
    A = _matrix(ctx)
    t = A.tile(0, 0)
    tp = pm.Dtd(ctx)
    events = []
    lock = threading.Lock()

    def ev(tag):
        with lock:
            events.append(tag)

    tp.insert_py(lambda: ev("w0"), flows=[(t, pm.ACCESS_OUT)])
    for i in range(8):
        tp.insert_py((lambda i=i: ev(("r", i))), flows=[(t, pm.ACCESS_IN)])
    tp.insert_py(lambda: ev("w1"), flows=[(t, pm.ACCESS_INOUT)])
    tp.wait()
    assert events[0] == "w0"
    assert events[-1] == "w1"
    assert {e for e in events[1:-1]} == {("r", i) for i in range(8)}


def test_independent_tiles_parallel(ctx):
    """Tasks on disjoint tiles are unordered (and actually overlap)."""
    A = _matrix(ctx, nt=2)
    tp = pm.Dtd(ctx)
    running = []
    peak = [0]
    lock = threading.Lock()
    import time

    def body():
        with lock:
            running.append(1)
            peak[0] = max(peak[0], len(running))
        time.sleep(0.05)
        with lock:
            running.pop()

    for tm in range(2):
        for tn in range(2):
            tp.insert_py(body, flows=[(A.tile(tm, tn), pm.ACCESS_INOUT)])
    tp.wait()
    assert peak[0] >= 2  # 2 workers in the fixture


def test_no_flow_tasks(ctx):
    tp = pm.Dtd(ctx)
    n = [0]
    lock = threading.Lock()

    def body():
        with lock:
            n[0] += 1

    for _ in range(100):
        tp.insert_py(body)
    tp.wait()
    assert n[0] == 100


def test_taskpool_reuse_after_wait(ctx):
    A = _matrix(ctx)
    t = A.tile(0, 0)
    tp = pm.Dtd(ctx)
    log = []
    tp.insert_py(lambda: log.append(1), flows=[(t, pm.ACCESS_INOUT)])
    tp.wait()
    tp.insert_py(lambda: log.append(2), flows=[(t, pm.ACCESS_INOUT)])
    tp.wait()
    assert log == [1, 2]


def test_sched_variants():
    """MCA sched module analog (mca/sched/*): every module (and every
    reference-name alias: lfq/gd/ll/ap/ltq/llp/lhq) runs the same DAG
    correctly — ws (lfq), fifo (gd), lifo (ll), spq (ap), rnd, pbq
    (ltq/llp/lhq: per-worker priority heaps + steal), ip (inverse
    priority)."""
    import subprocess, sys, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for kind in ["fifo", "lifo", "ws", "spq", "rnd", "pbq", "ip", "ltq",
                 "gd", "ap"]:
        code = f"""
import sys; sys.path.insert(0, {repo!r})
import numpy as np
import parsec_amd as pm
pm.param_set("sched", {kind!r})
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 42); pm.insert_potrf(tp, A); tp.wait()
d = A.tile_numpy(3, 3)
assert np.isfinite(d).all() and d[0,0] > 0
del A, ctx
"""
        r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                           text=True, timeout=120)
        assert r.returncode == 0, f"sched={kind}: {r.stdout}{r.stderr}"


def test_random_dag_vs_sequential_oracle(ctx):
    """Fuzz: a random program of scale/add/copy ops over 8 tiles executed
    through the DTD engine must match a sequential numpy replay (pins the
    RAW/WAR/WAW chaining semantics under concurrent execution)."""
    import numpy as np
    import random
    rng = random.Random(1234)
    nb = 32
    NT = 8
    mats = [pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1) for _ in range(NT)]
    ref = []
    tp = pm.Dtd(ctx)
    for i, M in enumerate(mats):
        v = np.full((nb, nb), float(i + 1))
        M.tile_numpy_set(0, 0, v)
        ref.append(v.copy())
    for _ in range(200):
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
    for i, M in enumerate(mats):
        got = M.tile_numpy(0, 0)
        assert np.allclose(got, ref[i]), f"tile {i} diverged"


def test_param_precedence():
    """MCA param sourcing: explicit set > env > default."""
    import subprocess, sys, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = f"""
import sys, os
sys.path.insert(0, {repo!r})
import parsec_amd as pm
# env already set by parent; param_set must override it
pm.param_set("dtd_window_size", "123")
ctx = pm.Context(nworkers=1, rank=0, world=1, gpu=-2)
tp = pm.Dtd(ctx)   # registers dtd params
dump = pm.param_dump()
assert "dtd_window_size" in dump
print("PARAM_OK")
del tp, ctx
"""
    env = dict(os.environ)
    env["PARSEC_MCA_dtd_window_size"] = "999"
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=60)
    assert "PARAM_OK" in r.stdout, r.stdout + r.stderr


def test_py_task_with_data(ctx):
    """Python bodies receive tile buffers (reference DTD bodies get tile
    pointers): a numpy-implemented DAG over real tile data."""
    import numpy as np
    nb = 32
    A = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    B = pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1)
    tp = pm.Dtd(ctx)

    def fill(buf):
        np.frombuffer(buf, dtype=np.float64)[:] = 3.0

    def double_into(src, dst):
        d = np.frombuffer(dst, dtype=np.float64)
        d[:] = np.frombuffer(src, dtype=np.float64) * 2.0

    def add_one(buf):
        np.frombuffer(buf, dtype=np.float64)[:] += 1.0

    tp.insert_py(fill, flows=[(A.tile(0, 0), pm.ACCESS_OUT)], with_data=True)
    tp.insert_py(double_into, flows=[(A.tile(0, 0), pm.ACCESS_IN),
                                     (B.tile(0, 0), pm.ACCESS_OUT)],
                 with_data=True)
    tp.insert_py(add_one, flows=[(B.tile(0, 0), pm.ACCESS_INOUT)],
                 with_data=True)
    tp.wait()
    assert np.allclose(B.tile_numpy(0, 0), 7.0)


def test_untied_tasks(ctx):
    """Tasks inserting tasks into the same taskpool from worker threads
    (dtd_test_untie analog) — insertion is internally serialized."""
    import threading
    A = pm.TiledMatrix(ctx, 8, 8, 1, 1, 1, 1)
    tp = pm.Dtd(ctx, "untie")
    hits = []
    lock = threading.Lock()

    def leaf(i):
        with lock:
            hits.append(i)

    def root():
        # runs on a worker thread; inserts 16 more tasks
        for i in range(16):
            tp.insert_py((lambda i=i: leaf(i)),
                         flows=[(A.tile(i % 8, i // 8), pm.ACCESS_INOUT)])

    tp.insert_py(root, flows=[(A.tile(0, 0), pm.ACCESS_INOUT)])
    tp.wait()
    assert sorted(hits) == list(range(16)), hits


def test_fatal_handler():
    """Error callback fires with the message before abort (weaksym_exit)."""
    import subprocess
    import sys as _sys
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.set_fatal_handler(lambda msg: print("HANDLER_SAW:", msg, flush=True))
ctx = pm.Context(nworkers=1, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 64, 64, 32, 32, 1, 1)
A.tile(5, 5)  # out of range -> fatal
"""
    r = subprocess.run([_sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode != 0
    assert "HANDLER_SAW:" in r.stdout, r.stdout + r.stderr


def test_grid_mismatch_rejected():
    """p*q != world is a loud error, not silent miscomputation."""
    import subprocess
    import sys as _sys
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
ctx = pm.Context(nworkers=1, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 64, 64, 32, 32, 2, 2)  # p*q=4 != world=1
"""
    r = subprocess.run([_sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode != 0 and "grid p*q" in (r.stderr + r.stdout)


def test_subtile_shape_mismatch_rejected():
    import subprocess
    import sys as _sys
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
ctx = pm.Context(nworkers=1, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
S = pm.TiledMatrix(ctx, 32, 32, 16, 16, 1, 1)  # wrong global shape
tp = pm.Dtd(ctx)
pm.insert_subtile_extract(tp, A, 0, 0, S)
"""
    r = subprocess.run([_sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode != 0 and "subtile" in (r.stderr + r.stdout)


def test_window_throttles_inserter():
    """dtd_window_size bounds how far insertion runs ahead of execution
    (insert_function.c:75-76 analog): with a small window and slow
    bodies, the insert loop itself takes execution-scale time."""
    import subprocess
    import sys as _sys
    code = f"""
import sys, time; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("dtd_window_size", "20")
pm.param_set("dtd_threshold_size", "10")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 8, 8, 8, 8, 1, 1)
tp = pm.Dtd(ctx)

def slow():
    time.sleep(0.002)

t0 = time.perf_counter()
for i in range(300):
    tp.insert_py(slow, flows=[(A.tile(0, 0), pm.ACCESS_INOUT)])
dt_insert = time.perf_counter() - t0
tp.wait()
# 300 serial 2ms bodies ~ 0.6s; a 20-deep window forces the inserter to
# ride along for most of it. Unthrottled insertion would take ~ms.
assert dt_insert > 0.2, dt_insert
print("WINDOW_OK", round(dt_insert, 3))
del A, ctx
"""
    r = subprocess.run([_sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "WINDOW_OK" in r.stdout, r.stdout + r.stderr


def test_zone_alloc_coalescing():
    """Slab allocator (zone_malloc analog): best-fit + neighbor merging.
    Freeing mixed-size interleaved blocks must restore one maximal block;
    round-1 size-class freelists failed this (capacity stranding)."""
    z = pm.ZoneAlloc(1 << 20)
    assert z.capacity == 1 << 20
    offs = [(z.alloc(sz), sz) for sz in (4096, 1024, 65536, 256, 4096)]
    assert all(o is not None for o, _ in offs)
    assert z.in_use == sum(sz for _, sz in offs)
    # free in scrambled order; neighbors must merge back
    import random
    random.Random(5).shuffle(offs)
    for o, sz in offs:
        z.free(o, sz)
    assert z.in_use == 0
    assert z.free_blocks == 1
    assert z.largest_free == 1 << 20
    # a request larger than any freelist class but smaller than the merged
    # block must succeed
    assert z.alloc((1 << 20) - 64) is not None


def test_zone_alloc_fragmentation_reuse():
    """Alternating alloc/free of different sizes must not strand capacity:
    after freeing everything, a full-slab allocation succeeds."""
    z = pm.ZoneAlloc(1 << 16)
    live = []
    import random
    rng = random.Random(11)
    for step in range(2000):
        if live and (rng.random() < 0.5 or z.largest_free < 2048):
            o, sz = live.pop(rng.randrange(len(live)))
            z.free(o, sz)
        else:
            sz = rng.choice([256, 512, 768, 1024, 2048])
            o = z.alloc(sz)
            if o is not None:
                live.append((o, sz))
    for o, sz in live:
        z.free(o, sz)
    assert z.in_use == 0 and z.free_blocks == 1
    assert z.alloc(1 << 16) == 0


def test_scheduler_modules():
    """MCA sched module variants (mca/sched analogs): spq (one shared
    priority queue) and rnd (random victim order) run the same DAG to the
    same result."""
    import subprocess, sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for kind in ("spq", "rnd", "fifo", "lifo", "ws"):
        code = f"""
import sys; sys.path.insert(0, {repo!r})
import parsec_amd as pm
pm.param_set("sched", {kind!r})
ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
tp = pm.Dtd(ctx)
acc = []
for i in range(500):
    tp.insert_py(lambda i=i: acc.append(i))
tp.wait()
assert len(acc) == 500
print("SCHED_OK", {kind!r})
del ctx
"""
        r = subprocess.run([sys.executable, "-c", code],
                           capture_output=True, text=True, timeout=120)
        assert r.returncode == 0 and "SCHED_OK" in r.stdout, (
            kind, r.stdout + r.stderr)


def test_pins_extended_events():
    """The extended PINS lifecycle set (create/release_deps/steal...)
    fires through the callback chain (pins.h:26-55 17-event parity)."""
    ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
    seen = set()
    # CREATE=6, RELEASE_DEPS=7 (see src/pins.hpp)
    h = pm.pins_add(lambda ev, name, wk: seen.add(ev),
                    ["create", "release_deps", "schedule", "complete"])
    tp = pm.Dtd(ctx)
    A = pm.TiledMatrix(ctx, 32, 32, 32, 32, 1, 1)
    import numpy as np
    A.tile_numpy_set(0, 0, np.zeros((32, 32)))
    for _ in range(4):
        tp.insert_py(lambda x: None, [(A.tile(0, 0), pm.ACCESS_INOUT)],
                     with_data=True)
    tp.wait()
    pm.pins_remove(h)
    assert {"create", "release_deps", "schedule", "complete"} <= seen, seen
    del A, ctx


def test_info_registry():
    """Per-context info registry (class/info.c analog): components publish
    facts at attach; user keys coexist."""
    ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
    info = ctx.info()
    assert info["runtime.workers"] == "2"
    assert info["sched.kind"] == "ws"
    assert info["comm.kind"] == "null"
    ctx.info_set("app.phase", "warmup")
    assert ctx.info()["app.phase"] == "warmup"
    del ctx


def test_gemm_fp64_dag_cpu_numerics():
    """insert_gemm_fp64 (tiled NN C=A*B; the hipGraph-replay demo DAG)
    against NumPy, including partial edge tiles; idempotence: re-running
    the pool reproduces the same C (k==0 overwrites with beta=0)."""
    import numpy as np
    ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
    A = pm.TiledMatrix(ctx, 300, 150, 64, 64, 1, 1)
    B = pm.TiledMatrix(ctx, 150, 200, 64, 64, 1, 1)
    C = pm.TiledMatrix(ctx, 300, 200, 64, 64, 1, 1)
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

    ref = full(A) @ full(B)
    for _ in range(2):  # second pass checks idempotence
        tp2 = pm.Dtd(ctx)
        pm.insert_gemm_fp64(tp2, A, B, C)
        tp2.wait()
        err = abs(full(C) - ref).max() / abs(ref).max()
        assert err < 1e-13, err
    del A, B, C, ctx


def test_gpu_graph_capture_requires_gpu():
    ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
    tp = pm.Dtd(ctx)
    with pytest.raises(RuntimeError, match="GPU"):
        tp.capture_begin()
    del tp, ctx


def test_fuzz_oracle_under_new_schedulers():
    """The random-DAG oracle fuzz under the round-2 scheduler modules
    (pbq per-worker heaps + steal, ip inverse priority): scheduling order
    must never change results."""
    import subprocess, sys, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for kind in ["pbq", "ip"]:
        code = f"""
import sys, random; sys.path.insert(0, {repo!r})
import numpy as np
import parsec_amd as pm
pm.param_set("sched", {kind!r})
ctx = pm.Context(nworkers=4, rank=0, world=1, gpu=-2)
rng = random.Random(77)
nb, NT = 32, 6
mats = [pm.TiledMatrix(ctx, nb, nb, nb, nb, 1, 1) for _ in range(NT)]
ref = [np.full((nb, nb), float(i + 1)) for i in range(NT)]
tp = pm.Dtd(ctx)
for i, M in enumerate(mats):
    M.tile_numpy_set(0, 0, ref[i])
def sc(x, a, b):
    v = np.frombuffer(x, dtype=np.float64)
    v *= a; v += b
def ad(x, y):
    np.frombuffer(y, dtype=np.float64)[:] += np.frombuffer(x, np.float64)
for step in range(120):
    op = rng.randrange(2)
    i, j = rng.randrange(NT), rng.randrange(NT)
    prio = rng.randrange(-50, 50)
    if op == 0:
        a, b = rng.uniform(0.5, 1.5), rng.uniform(-1, 1)
        tp.insert_py(lambda x, _a=a, _b=b: sc(x, _a, _b),
                     [(mats[i].tile(0, 0), pm.ACCESS_INOUT)],
                     priority=prio, with_data=True)
        ref[i] = ref[i] * a + b
    elif i != j:
        tp.insert_py(lambda x, y: ad(x, y),
                     [(mats[i].tile(0, 0), pm.ACCESS_IN),
                      (mats[j].tile(0, 0), pm.ACCESS_INOUT)],
                     priority=prio, with_data=True)
        ref[j] = ref[j] + ref[i]
tp.wait()
for i, M in enumerate(mats):
    assert np.allclose(M.tile_numpy(0, 0), ref[i]), f"tile {{i}} diverged"
print("FUZZ_OK")
del mats, tp, ctx
"""
        r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                           text=True, timeout=300)
        assert r.returncode == 0 and "FUZZ_OK" in r.stdout, \
            f"sched={kind}: {r.stdout}{r.stderr}"


def test_untracked_flow_semantics():
    """ACCESS_UNTRACKED (PARSEC_DONT_TRACK analog): the flow is exempt
    from chaining — a later tracked writer must NOT WAR-serialize behind
    an untracked reader. Proven deterministically with an event
    handshake: the reader refuses to finish until the writer has run,
    which would deadlock-timeout if the engine added the WAR edge."""
    import numpy as np
    import threading
    ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
    A = pm.TiledMatrix(ctx, 128, 64, 64, 64, 1, 1)
    tp = pm.Dtd(ctx)
    A.tile_numpy_set(0, 0, np.full((64, 64), 7.0))
    wrote = threading.Event()
    overlapped = []

    def reader(buf):
        # with tracking, the writer would be WAR-blocked on this task and
        # this wait would time out
        overlapped.append(wrote.wait(timeout=20))

    def writer(buf):
        np.frombuffer(buf, dtype=np.float64)[:] = 9.0
        wrote.set()

    tp.insert_py(reader,
                 [(A.tile(0, 0), pm.ACCESS_IN | pm.ACCESS_UNTRACKED)],
                 with_data=True)
    tp.insert_py(writer, [(A.tile(0, 0), pm.ACCESS_OUT)], with_data=True)
    tp.wait()
    assert overlapped == [True], "writer serialized behind untracked reader"
    assert np.all(A.tile_numpy(0, 0) == 9.0)
    # untracked read correctness with NO concurrent writer
    got = []
    tp.insert_py(lambda b: got.append(np.frombuffer(b, np.float64)[0]),
                 [(A.tile(0, 0), pm.ACCESS_IN | pm.ACCESS_UNTRACKED)],
                 with_data=True)
    tp.wait()
    assert got == [9.0], got
    del A, tp, ctx


def test_empty_taskpool_wait(ctx):
    """dtd_test_empty analog: waiting an empty pool returns immediately;
    wait() is re-entrant after completion."""
    tp = pm.Dtd(ctx)
    tp.wait()
    tp.wait()
    tp.wait_dynamic()


def test_multiple_taskpool_concurrent_wait(ctx):
    """dtd_test_multiple_handle_wait analog: two pools driven from two
    application threads, each waiting its own, both complete."""
    import threading
    import numpy as np
    A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
    B = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
    errs = []

    def run(M, seed):
        try:
            tp = pm.Dtd(ctx)
            pm.insert_spd_fill(tp, M, seed)
            pm.insert_potrf(tp, M)
            tp.wait()
            assert np.isfinite(M.tile_numpy(3, 3)).all()
        except Exception as e:  # propagate to the main thread
            errs.append(e)

    t1 = threading.Thread(target=run, args=(A, 1))
    t2 = threading.Thread(target=run, args=(B, 2))
    t1.start(); t2.start(); t1.join(); t2.join()
    assert not errs, errs
    del A, B
