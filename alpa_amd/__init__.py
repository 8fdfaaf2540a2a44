"""alpa_amd — MI355X-native auto-parallelization framework.

A from-scratch, MI355X-first framework with the capabilities of Alpa
(alpa-projects/alpa): automatic intra-operator sharding (ILP), inter-operator
pipeline parallelism (DP stage slicing, 1F1B), ZeRO, MoE/expert parallelism,
checkpointing, and auto-sharded serving — built on PyTorch-ROCm +
hand-written CDNA4 (gfx950) HIP kernels + RCCL collectives over xGMI, one
process per GPU.  No JAX/XLA, no Ray, no CUDA shims.
"""

from .api import (TrainState, grad, init, parallelize, shutdown,
                  value_and_grad)
from .serialization import (restore_checkpoint, restore_train_state,
                            save_checkpoint, save_train_state)
from .global_env import global_config
from .mesh import (DeviceMesh, VirtualMesh, device, full_mesh, full_virtual_mesh,
                   get_device_mesh, init_distributed, local_rank, rank,
                   world_size)
from .optim import AdamW
from .shard_parallel import auto_shard, capture_graph  # noqa: F401
from .pipeline_parallel.boundary import (automatic_remat,  # noqa: F401
                                         manual_remat,
                                         mark_pipeline_boundary,
                                         spec_from_module)
from .parallel_method import (AutoShardingOption, CreateStateParallel,
                              DataParallel, FollowParallel, ParallelMethod,
                              PipeshardParallel, ShardParallel,
                              Zero2Parallel, Zero3Parallel,
                              get_3d_parallel_method, parallelize_inference)

from .version import __version__, check_hip_ops_version  # noqa: F401

__all__ = [
    "init", "shutdown", "parallelize", "TrainState", "AdamW",
    "DeviceMesh", "VirtualMesh", "device", "full_mesh", "full_virtual_mesh",
    "get_device_mesh", "init_distributed", "local_rank", "rank", "world_size",
    "ParallelMethod", "ShardParallel", "DataParallel", "Zero2Parallel",
    "Zero3Parallel", "PipeshardParallel", "AutoShardingOption",
    "CreateStateParallel", "FollowParallel", "parallelize_inference",
    "get_3d_parallel_method", "global_config", "grad", "value_and_grad",
    "save_checkpoint", "restore_checkpoint", "save_train_state",
    "restore_train_state",
]
