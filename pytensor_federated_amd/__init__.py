"""MI355X-native federated logp/gradient engine.

A brand-new framework with the capabilities of
``michaelosthege/pytensor-federated`` (reference layer map: SURVEY.md §1),
designed MI355X-first:

* transport (L0-L3): hand-written proto3 codec + grpc.aio service, wire
  compatible with the reference's ``service.proto``;
* compute: PyTorch-ROCm models with hand-written CDNA4 HIP kernels for the
  Gaussian/GLM logp+grad hot path (``pytensor_federated_amd.ops``);
* on-node scale-out: 8 GPUs as 8 federated shards, per-shard
  ``[logp, grads]`` summed via RCCL all-reduce over xGMI
  (``pytensor_federated_amd.parallel``);
* graph embedding: ``torch.autograd``-native LogpGradOp
  (``pytensor_federated_amd.torch_ops``), an async task-graph engine with
  automatic fan-out fusion (``op_async``), and optional PyTensor adapters
  (``wrapper_ops``) when pytensor is installed.

Like the reference's ``__init__.py:1-12``, graph-layer imports are optional
so the transport stack works standalone.
"""
import os as _os

# grpc's default epoll1 poller aborts forked children; these must be set
# before the first `import grpc` anywhere in the process.  Needed for parity
# with the reference's fork-based multiprocessing support (PyMC chain
# workers re-connect after fork via the `thread_pid_id` keying).
_os.environ.setdefault("GRPC_ENABLE_FORK_SUPPORT", "true")
_os.environ.setdefault("GRPC_POLL_STRATEGY", "epoll1")

from . import npproto, rpc  # noqa: F401
from .common import (  # noqa: F401
    LogpGradServiceClient,
    LogpServiceClient,
    wrap_logp_func,
    wrap_logp_grad_func,
)
from .op_async import (  # noqa: F401
    AsyncComputeNode,
    AsyncTaskGraph,
    fuse_parallel_layers,
    gather_evaluate,
)
from .service import (  # noqa: F401
    ArraysToArraysService,
    ArraysToArraysServiceClient,
    get_load_async,
    get_loads_async,
)
from .signatures import ComputeFunc, LogpFunc, LogpGradFunc  # noqa: F401

try:
    from . import inference, models, parallel  # noqa: F401
    from .torch_ops import LogpGradOp, LogpOp, federated_logp_grad  # noqa: F401
except ModuleNotFoundError:
    # torch not installed: transport-only deployment (e.g. a CPU client box).
    pass

try:
    # PyTensor graph adapters -- only when the user has pytensor installed
    # (parity with the reference's optional L4/L5 import, __init__.py:1-12).
    from .wrapper_ops import (  # noqa: F401
        AsyncLogpGradOp,
        AsyncLogpOp,
        LogpGradOp as PyTensorLogpGradOp,
        LogpOp as PyTensorLogpOp,
    )
except ModuleNotFoundError:
    pass

__version__ = "0.2.0"
