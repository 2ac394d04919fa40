"""`python -m parsec_amd` — version/config dump (--parsec-help analog)."""
import sys

import parsec_amd as pm


def main():
    if len(sys.argv) > 1 and sys.argv[1] == "bandwidth":
        # tools/gpu/testbandwidth analog: HIP link bandwidth probe
        if pm.hip_device_count() == 0:
            print("no HIP device visible"); return
        r = pm.hip_bandwidth()
        mb = r["bytes"] >> 20
        print(f"HIP copy bandwidth ({mb} MiB, pinned host):")
        print(f"  H2D {r['h2d_gbs']:.1f} GB/s   D2H {r['d2h_gbs']:.1f} GB/s"
              f"   D2D {r['d2d_gbs']:.1f} GB/s")
        return
    print(f"parsec_amd {pm.__version__} — MI355X-native task-dataflow "
          "runtime (PaRSEC-class)")
    print(f"HIP devices visible: {pm.hip_device_count()}")
    print("\nMCA-style parameters (set via PARSEC_MCA_<name> or "
          "parsec_amd.param_set):\n")
    # register the common params by touching a context-free surface
    print(pm.param_dump())
    print("Key params: sched_workers, "
          "sched (ws|fifo|lifo|spq|pbq|ip|rnd + reference-name aliases), "
          "sched_bind (0|1|numa), gpu_exec_streams, gpu_max_inflight, "
          "gpu_mem_percent, gpu_mem_limit_mb, chore_gemm (rocblas|hip), "
          "chore_syrk (hip|dgemm|syrkx), chore_potrf (hip|rocsolver), "
          "chore_qr (hand|rocsolver), qr_algo (house|bcgs), trsm_variant "
          "(invgemm|rocblas), dtd_window_size, comm_kind (tcp|rccl), "
          "comm_max_inflight, comm_send_reserve, bcast_tree "
          "(unicast|binomial), comm_base_port, profile_filename, "
          "profile_dot, profile_roctx, stats, debug_history, "
          "pins (task_profiler,print_steals,iterators_checker), "
          "live_stats, live_stats_interval_ms, graph_debug")
    print("\nTools: python -m parsec_amd.tools.trace2chrome | "
          "trace2pandas | live_top ;  python -m parsec_amd.ptg file.jdf")


if __name__ == "__main__":
    main()
