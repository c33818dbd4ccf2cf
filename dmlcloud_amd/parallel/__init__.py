"""Distributed runtime: bootstrap, collectives, DP strategies, hipGraphs."""

from .distributed import (  # noqa: F401
    all_gather_object,
    barrier,
    broadcast_object,
    deinitialize_torch_distributed,
    gather_object,
    has_environment,
    has_mpi,
    has_slurm,
    init_process_group_MPI,
    init_process_group_auto,
    init_process_group_dummy,
    init_process_group_env,
    init_process_group_slurm,
    is_root,
    local_node,
    local_rank,
    local_world_size,
    new_gloo_group,
    print_root,
    print_worker,
    rank,
    root_first,
    root_only,
    world_size,
)
from .ddp import XGMI_BUCKET_CAP_MB, wrap_ddp  # noqa: F401
from .flat import FlatAdam, FlatOptimizer, FlatReplica, FlatSGD  # noqa: F401
from .graphs import GraphedStep  # noqa: F401
