"""`python -m parsec_amd` — version/config dump (--parsec-help analog)."""
import sys

import parsec_amd as pm


def main():
    print(f"parsec_amd {pm.__version__} — MI355X-native task-dataflow "
          "runtime (PaRSEC-class)")
    print(f"HIP devices visible: {pm.hip_device_count()}")
    print("\nMCA-style parameters (set via PARSEC_MCA_<name> or "
          "parsec_amd.param_set):\n")
    # register the common params by touching a context-free surface
    print(pm.param_dump())
    print("Key params: sched_workers, sched (ws|fifo|lifo), "
          "gpu_exec_streams, gpu_max_inflight, gpu_mem_percent, "
          "gpu_mem_limit_mb, chore_gemm (rocblas|hip), chore_syrk "
          "(hip|dgemm|syrkx), chore_potrf (hip|rocsolver), trsm_variant "
          "(invgemm|rocblas), dtd_window_size, comm_kind (tcp|rccl), "
          "comm_base_port, profile_filename, profile_dot, stats")


if __name__ == "__main__":
    main()
