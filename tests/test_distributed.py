"""Multi-process distributed tests on the CPU TCP comm engine.

Exercises the deterministic SPMD dataflow protocol (sends/recvs, channel
sequencing, flush_all, barrier) with world_size 2 and 4 on one host —
the reference tests multi-node the same way (multiple ranks on one host,
SURVEY.md §4).
"""
import os
import subprocess
import sys

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
WORKER = os.path.join(HERE, "_dist_worker.py")

from conftest import port_base

_next_port = [port_base()]


def run_world(world, tmpdir, p, q, n=256, nb=64):
    port = _next_port[0]
    _next_port[0] += world + 2
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE=str(world),
                   PARSEC_TEST_PORT=str(port), PARSEC_TEST_OUT=str(tmpdir),
                   GRID_P=str(p), GRID_Q=str(q), MAT_N=str(n), MAT_NB=str(nb))
        procs.append(subprocess.Popen([sys.executable, WORKER], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for pr in procs:
        out, _ = pr.communicate(timeout=180)
        outs.append(out.decode())
        assert pr.returncode == 0, f"worker failed:\n{out.decode()}"
    # assemble
    pre = np.zeros((n, n))
    post = np.zeros((n, n))
    for r in range(world):
        z = np.load(os.path.join(tmpdir, f"rank{r}.npz"))
        for key in z.files:
            kind, tm, tn = key.split("_")[0], *key.split("_")[1:]
            tm, tn = int(tm), int(tn)
            v = z[key]
            dst = pre if kind == "pre" else post
            dst[tm * nb:tm * nb + v.shape[0], tn * nb:tn * nb + v.shape[1]] = v
    return pre, post


@pytest.mark.parametrize("world,p,q", [(2, 2, 1), (2, 1, 2), (4, 2, 2), (8, 2, 4)])
def test_distributed_cholesky(world, p, q, tmp_path):
    pre, post = run_world(world, tmp_path, p, q)
    M = np.tril(pre) + np.tril(pre, -1).T
    L0 = np.linalg.cholesky(M)
    err = np.abs(np.tril(post) - L0).max()
    assert err < 1e-10, f"world={world} p={p} q={q}: max err {err}"


def test_distributed_qr(tmp_path):
    """World-2 tile QR over the TCP engine: checks workspace-collection
    transfers (V2/T1 tiles) and flush; verified via R^T R == A^T A."""
    world, p, q, n, nb = 2, 2, 1, 256, 64
    port = _next_port[0]
    _next_port[0] += world + 2
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update(RANK=str(r), WORLD_SIZE=str(world),
                   PARSEC_TEST_PORT=str(port), PARSEC_TEST_OUT=str(tmp_path),
                   GRID_P=str(p), GRID_Q=str(q), MAT_N=str(n), MAT_NB=str(nb),
                   PARSEC_TEST_APP="qr")
        procs.append(subprocess.Popen([sys.executable, WORKER], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for pr in procs:
        out, _ = pr.communicate(timeout=180)
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
    R = np.triu(post)
    err = np.abs(R.T @ R - pre.T @ pre).max() / np.abs(pre.T @ pre).max()
    assert err < 1e-12, f"distributed QR rel err {err}"
