#!/usr/bin/env python3
"""PMC harness: run one fp64 GEMM variant under rocprofv3 --pmc.
argv[1]: hand | rocblas"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from parsec_amd._core import bench_dgemm_hip, bench_dgemm_rocblas  # noqa: E402

which = sys.argv[1] if len(sys.argv) > 1 else "hand"
n, iters = 4096, 5
if which == "hand":
    dt = bench_dgemm_hip(n, n, n, iters, 0)
else:
    dt = bench_dgemm_rocblas(n, n, n, iters)
print(f"{which}: {2 * n**3 * iters / dt / 1e12:.1f} TF")
