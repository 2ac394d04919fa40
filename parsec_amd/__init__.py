"""parsec_amd: MI355X-native task-dataflow runtime (PaRSEC-class).

A from-scratch DAG task runtime for one AMD MI355X node (8x CDNA4/gfx950,
288 GB HBM3E each, xGMI mesh): PaRSEC-style taskpool/DTD programming model,
native HIP device engine (hipStream/hipEvent pipeline, pooled HBM), RCCL
dataflow communication between one-process-per-GPU ranks, and hand-written
CDNA4 MFMA tile kernels for the dense linear-algebra headline apps.

See SURVEY.md for the structural map of the reference (ICLDisco/parsec)
this framework re-implements natively.
"""

import os
import sys

from parsec_amd._core import (  # noqa: F401,E402
    ACCESS_IN,
    ACCESS_INOUT,
    ACCESS_OUT,
    Context,
    Data,
    Dtd,
    IrregularCollection,
    Taskpool,
    TiledMatrix,
    hip_device_count,
    insert_potrf,
    insert_geqrf,
    insert_fill_bf16,
    insert_gemm_bf16,
    insert_spd_fill,
    insert_full_fill,
    insert_redistribute,
    insert_apply_scale,
    insert_reduce_sum,
    insert_stencil_1d,
    insert_panel_fill,
    insert_potrf_panel,
    param_dump,
    param_set,
    hip_bandwidth,
    pins_add,
    pins_remove,
    set_fatal_handler,
)

__version__ = "0.1.0"


def _env_int(name, default):
    v = os.environ.get(name)
    return int(v) if v else default


def init_distributed(nworkers=-1, comm=None, gpu=-1):
    """Create a Context for this rank.

    Under torch.distributed.run (one process per GPU over RCCL), rank/world
    come from RANK/WORLD_SIZE. The RCCL bootstrap (ncclUniqueId) is exchanged
    through torch.distributed's gloo store when world > 1 and GPUs are
    present; otherwise the TCP host engine is used (CPU paths / tests).
    """
    from parsec_amd import _core

    rank = _env_int("RANK", 0)
    world = _env_int("WORLD_SIZE", 1)
    if comm is None:
        comm = os.environ.get("PARSEC_COMM_KIND")
    if comm is None:
        comm = "rccl" if (world > 1 and hip_device_count() > 0) else ""
    if world > 1:
        # gloo side-channel for bootstrap + whole-job reductions (bench
        # timing); works with or without GPUs.
        import torch.distributed as dist

        if not dist.is_initialized():
            dist.init_process_group(backend="gloo", rank=rank, world_size=world)
        if comm == "rccl":
            if rank == 0:
                obj = [_core.nccl_unique_id()]
            else:
                obj = [None]
            dist.broadcast_object_list(obj, src=0)
            _core.set_nccl_unique_id(obj[0])
    return Context(nworkers=nworkers, rank=rank, world=world, comm=comm, gpu=gpu)
