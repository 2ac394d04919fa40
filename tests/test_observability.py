"""Tracing, DOT grapher, stats counters (SURVEY.md §5 aux subsystems)."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_trace_and_dot(tmp_path):
    trace = tmp_path / "trace"
    dot = tmp_path / "dag.dot"
    code = f"""
import sys
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("profile_filename", {str(trace)!r})
pm.param_set("profile_dot", {str(dot)!r})
pm.param_set("stats", "1")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
del A
del ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "[parsec_amd stats]" in r.stderr
    tfile = str(trace) + ".0"
    assert os.path.exists(tfile), "trace file missing"
    from parsec_amd.tools.trace2chrome import convert
    out, n = convert(tfile, str(tmp_path / "t.json"))
    assert n > 5, f"too few trace events: {n}"
    dfile = str(dot) + ".0"
    dtext = open(dfile).read()
    assert "digraph" in dtext
    assert "potrf" in dtext
    assert "->" in dtext


def test_counters_and_cli(tmp_path):
    import parsec_amd as pm
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 1); tp.wait()
c = ctx.counters()
assert c["tasks_executed_cpu"] >= 3, c
assert c["tasks_scheduled"] >= 3, c
print("COUNTERS_OK")
del A, tp, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "COUNTERS_OK" in r.stdout, r.stdout + r.stderr
    r2 = subprocess.run([sys.executable, "-m", "parsec_amd"],
                        capture_output=True, text=True, timeout=120, cwd=REPO)
    assert "MI355X-native" in r2.stdout


def test_roctx_sink_cpu_noop(tmp_path):
    """profile_roctx=1 wraps bodies in rocTX ranges; harmless w/o rocprof."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("profile_roctx", "1")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
print("ROCTX_OK")
del A, tp, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "ROCTX_OK" in r.stdout, r.stdout + r.stderr


def test_pins_callbacks():
    """PINS chain (mca/pins analog): lifecycle callbacks with event mask."""
    code = f"""
import sys, threading; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
seen = {{}}
lock = threading.Lock()
def cb(ev, cls, worker):
    with lock:
        seen[ev] = seen.get(ev, 0) + 1
h = pm.pins_add(cb, ["exec_begin", "exec_end", "schedule", "complete"])
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
assert seen["exec_begin"] == seen["exec_end"] >= 10, seen
assert seen["complete"] >= seen["exec_begin"], seen
assert seen["schedule"] >= 10, seen
pm.pins_remove(h)
n0 = dict(seen)
tp2 = pm.Dtd(ctx)
pm.insert_potrf(tp2, A)
tp2.wait()
assert seen == n0, "callbacks fired after removal"
print("PINS_OK")
del A, tp, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "PINS_OK" in r.stdout, r.stdout + r.stderr


def test_debug_history_on_fatal():
    """PARSEC_MCA_debug_history=N dumps the event ring on fatal (debug
    history analog)."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("debug_history", "64")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 128, 128, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
tp.wait()
A.tile(9, 9)  # out of range -> fatal -> history dump
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode != 0
    assert "debug history" in r.stderr and "spd_fill" in r.stderr, r.stderr


def test_tutorial_snippets_execute():
    """The tutorial's code blocks stay runnable (doc-rot guard)."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
ctx = pm.Context(nworkers=2, gpu=-2)
A = pm.TiledMatrix(ctx, 512, 512, 128, 128, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, seed=42)
pm.insert_potrf(tp, A)
tp.insert_py(lambda: None, flows=[(A.tile(0, 0), pm.ACCESS_IN)])
tp.wait()
tp.flush_all(A)
from parsec_amd.ptg import compile_jdf
import os as _os
mod = compile_jdf(_os.path.join({REPO!r}, "examples", "cholesky.jdf"))
tp2 = pm.Dtd(ctx)
mod.build(ctx, tp2, descA=A, NT=A.mt, NB=A.nb)
tp2.wait()
# round-2 features: reshape promise + wait_dynamic (world-1 degenerates
# to wait)
import numpy as np
R = pm.TiledMatrix(ctx, 128, 128, 128, 128, 1, 1)
tp3 = pm.Dtd(ctx)
def tr(x, out):
    np.frombuffer(out, dtype=np.float64)[:] = np.frombuffer(x, dtype=np.float64)
tp3.insert_py(tr, [(A.tile(0, 0), pm.ACCESS_IN, pm.RESHAPE_TRANSPOSE),
                   (R.tile(0, 0), pm.ACCESS_OUT)], with_data=True)
tp3.wait_dynamic()
print("SNIPPETS_OK")
del A, R, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300)
    assert "SNIPPETS_OK" in r.stdout, r.stdout + r.stderr


def test_comm_trace_events(tmp_path):
    """Comm send/recv land in the binary trace (remote_dep.h:384-419
    comm-event tracing analog)."""
    from conftest import port_base
    code = f"""
import os, sys
sys.path.insert(0, {REPO!r})
import parsec_amd as pm
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
pm.param_set("profile_filename", os.environ["TRACE"])
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 2, 1)
tp = pm.Dtd(ctx, "ct")
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
ctx.barrier()
del A, tp, ctx
print("CT_OK", rank)
"""
    port = str(port_base(25))
    trace = str(tmp_path / "ctrace")
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port, TRACE=trace)
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        o, _ = pr.communicate(timeout=180)
        assert pr.returncode == 0 and b"CT_OK" in o, o.decode()
    from parsec_amd.tools.trace2chrome import convert
    found_comm = 0
    for r in range(2):
        out, n = convert(f"{trace}.{r}", str(tmp_path / f"t{r}.json"))
        text = open(str(tmp_path / f"t{r}.json")).read()
        found_comm += text.count("comm_send") + text.count("comm_recv")
    assert found_comm > 0, "no comm events in either rank's trace"


def test_pins_builtin_modules():
    """PARSEC_MCA_pins=task_profiler,print_steals,iterators_checker: the
    built-in PINS modules (mca/pins/* analogs) report at teardown."""
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("pins", "task_profiler,print_steals,iterators_checker")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 512, 512, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
del A, tp, ctx
print("MODS_OK")
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "MODS_OK" in r.stdout, r.stdout + r.stderr
    assert "[pins:task_profiler]" in r.stderr and "potrf" in r.stderr, r.stderr
    assert "[pins:print_steals]" in r.stderr
    assert "[pins:iterators_checker]" in r.stderr
    assert "OK: every created task completed" in r.stderr, r.stderr
    assert "ANOMALY" not in r.stderr


def test_jacobi_replay_example_cpu():
    """examples/jacobi_replay.py converges on the CPU engine path."""
    r = subprocess.run([sys.executable,
                        os.path.join(REPO, "examples", "jacobi_replay.py"),
                        "3", "32", "40"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "fixed-point err" in r.stdout


def test_live_stats_and_live_top(tmp_path):
    """PARSEC_MCA_live_stats publishes periodic JSON snapshots (properties
    dictionary / aggregator_visu analog) and live_top renders them."""
    live = str(tmp_path / "live")
    code = f"""
import sys, time; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("live_stats", {live!r})
pm.param_set("live_stats_interval_ms", "50")
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 512, 512, 64, 64, 1, 1)
for _ in range(4):
    tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 1); pm.insert_potrf(tp, A)
    tp.wait()
time.sleep(0.3)  # let at least one snapshot land after the work
import json
s = json.load(open({live!r} + ".0"))
assert s["tasks_cpu"] + s["tasks_gpu"] > 50, s
assert s["workers"] == 2 and s["rank"] == 0, s
print("LIVE_OK")
del A, tp, ctx
import os
assert not os.path.exists({live!r} + ".0"), "snapshot not removed at fini"
print("CLEAN_OK")
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert "LIVE_OK" in r.stdout and "CLEAN_OK" in r.stdout, \
        r.stdout + r.stderr
    # live_top --once renders a table from a snapshot file
    (tmp_path / "live2.0").write_text(
        '{"rank": 0, "world": 1, "workers": 2, "uptime_s": 1.5, '
        '"ready_queue": 3, "tasks_cpu": 100, "tasks_gpu": 5, '
        '"scheduled": 105, "steals": 2, "comm_msgs": 0, "comm_bytes": 0, '
        '"renames": 0}\n')
    r2 = subprocess.run([sys.executable, "-m", "parsec_amd.tools.live_top",
                         str(tmp_path / "live2"), "--once"],
                        capture_output=True, text=True, timeout=60, cwd=REPO)
    assert r2.returncode == 0, r2.stdout + r2.stderr
    assert "rank" in r2.stdout and "100" in r2.stdout, r2.stdout


def test_trace2pandas(tmp_path):
    """PABT1 -> pandas trace tables (pbt2ptt analog): load + summary."""
    trace = tmp_path / "ptrace"
    code = f"""
import sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
pm.param_set("profile_filename", {str(trace)!r})
ctx = pm.Context(nworkers=2, rank=0, world=1, gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 1, 1)
tp = pm.Dtd(ctx)
pm.insert_spd_fill(tp, A, 1)
pm.insert_potrf(tp, A)
tp.wait()
del A, tp, ctx
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    from parsec_amd.tools.trace2pandas import load, summarize
    df = load([str(trace) + ".0"])
    assert len(df) > 10
    assert {"kind", "class", "dur_us", "lane"} <= set(df.columns)
    assert (df["class"] == "potrf").any()
    s = summarize(df)
    assert s["count"].sum() == len(df)
    out = tmp_path / "t.csv"
    r2 = subprocess.run([sys.executable, "-m",
                         "parsec_amd.tools.trace2pandas",
                         str(trace) + ".0", "--out", str(out)],
                        capture_output=True, text=True, timeout=120,
                        cwd=REPO)
    assert r2.returncode == 0 and out.exists(), r2.stdout + r2.stderr


def test_live_stats_multirank(tmp_path):
    """Per-rank live snapshots: two ranks publish to <path>.<rank> and
    live_top aggregates both."""
    from conftest import port_base
    live = str(tmp_path / "mlive")
    code = f"""
import os, sys, time; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
pm.param_set("live_stats", {live!r})
pm.param_set("live_stats_interval_ms", "50")
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 2, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 1); pm.insert_potrf(tp, A)
tp.wait()
time.sleep(0.3)
import json
s = json.load(open({live!r} + f".{{rank}}"))
assert s["rank"] == rank and s["world"] == 2, s
ctx.barrier()
print("MLIVE_OK", rank)
del A, tp, ctx
"""
    port = str(port_base(35))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        out, _ = pr.communicate(timeout=120)
        assert pr.returncode == 0 and b"MLIVE_OK" in out, out.decode()


def test_trace2pandas_multirank(tmp_path):
    """trace tables merge multiple rank files with a rank column."""
    from conftest import port_base
    trace = str(tmp_path / "mtrace")
    code = f"""
import os, sys; sys.path.insert(0, {REPO!r})
import parsec_amd as pm
rank = int(os.environ["RANK"])
pm.param_set("comm_base_port", os.environ["PORT"])
pm.param_set("profile_filename", {trace!r})
ctx = pm.Context(nworkers=2, rank=rank, world=2, comm="tcp", gpu=-2)
A = pm.TiledMatrix(ctx, 256, 256, 64, 64, 2, 1)
tp = pm.Dtd(ctx); pm.insert_spd_fill(tp, A, 1); pm.insert_potrf(tp, A)
tp.wait()
ctx.barrier()
del A, tp, ctx
print("MT_OK", rank)
"""
    port = str(port_base(37))
    procs = []
    for r in range(2):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE="2", PORT=port)
        procs.append(subprocess.Popen([sys.executable, "-c", code], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        out, _ = pr.communicate(timeout=120)
        assert pr.returncode == 0 and b"MT_OK" in out, out.decode()
    from parsec_amd.tools.trace2pandas import load
    df = load([f"{trace}.0", f"{trace}.1"])
    assert set(df["rank"].unique()) == {0, 1}
    assert (df["kind"] == "comm_send").any() or \
        (df["kind"] == "comm_recv").any()
    # multi-rank Chrome merge: one timeline, rank r under pid 10r
    from parsec_amd.tools.trace2chrome import convert_many
    import json as _json
    out, n = convert_many([f"{trace}.0", f"{trace}.1"],
                          str(tmp_path / "merged.json"))
    evs = _json.load(open(out))["traceEvents"]
    assert n > 10 and {e.get("pid") for e in evs} >= {0, 10}
