"""Parallel execution: RCCL shard-DP across GPUs, HIP streams within one.

See SURVEY.md §2.4: the reference's parallelism axes map to
* worker-pool data parallelism  -> :class:`FederatedShardEngine` (RCCL/xGMI)
* graph-level async fan-out     -> :class:`MultiShardDispatcher` (HIP streams)
  and the asyncio fan-out in ``op_async`` for off-node workers.
"""
from .collective import FederatedShardEngine, init_process_group_from_env, shard_slice  # noqa: F401
from .streams import MultiShardDispatcher, NativeMultiShardEngine  # noqa: F401
