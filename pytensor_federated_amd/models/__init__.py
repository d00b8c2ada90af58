"""Worker-side model families (private-data logp+grad blackboxes)."""
from .base import LogpGradModel  # noqa: F401
from .linear import GaussianLinearModel, generate_linear_dataset  # noqa: F401
from .logistic import LogisticGLMModel, generate_logistic_dataset  # noqa: F401
from .ode import ODEModel, generate_ode_dataset, lotka_volterra_rhs  # noqa: F401
