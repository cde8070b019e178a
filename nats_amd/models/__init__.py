from .init import ortho_weight, norm_weight, init_params, PARAM_SCHEMA, param_shapes
from .distraction import NatsModel, default_options

__all__ = [
    "ortho_weight",
    "norm_weight",
    "init_params",
    "PARAM_SCHEMA",
    "param_shapes",
    "NatsModel",
    "default_options",
]
