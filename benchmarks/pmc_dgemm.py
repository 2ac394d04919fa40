#!/usr/bin/env python3
"""PMC harness: run one fp64 GEMM variant under rocprofv3 --pmc.
argv[1]: hand | rocblas"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from parsec_amd._core import bench_dgemm  # noqa: E402

which = sys.argv[1] if len(sys.argv) > 1 else "hand"
n, iters = 4096, 5
if which == "bf16":
    from parsec_amd._core import bench_gemm_bf16
    print(f"bf16: {bench_gemm_bf16(n, n, n, 20):.0f} TF")
else:
    tf = bench_dgemm(n, n, n, iters, "rocblas" if which == "rocblas" else "v2")
    print(f"{which}: {tf:.1f} TF")
