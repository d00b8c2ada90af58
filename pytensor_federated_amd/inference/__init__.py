"""Self-contained inference drivers over the LogpGradFunc contract.

The reference delegates sampling to PyMC (demo_model.py:38-44 runs
``pm.find_MAP`` + ``pm.sample``).  This framework works under PyMC too (via
``wrapper_ops``), but also ships its own drivers so a deployment needs
nothing beyond this package: every driver consumes the framework's
``LogpFunc`` / ``LogpGradFunc`` signatures -- a local model, a remote
``LogpGradServiceClient.evaluate``, or a ``FederatedShardEngine`` all plug
in unchanged (the logp+grad of one MCMC step is ONE fused worker call).
"""
from .diagnostics import effective_sample_size, split_rhat, summary  # noqa: F401
from .mala import sample_mala_batched  # noqa: F401
from .map import find_map  # noqa: F401
from .mcmc import Metropolis, sample_metropolis  # noqa: F401
from .nuts import NUTS, sample_nuts  # noqa: F401
from .nuts_batched import sample_nuts_batched  # noqa: F401
