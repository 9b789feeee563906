from . import functional  # noqa: F401
from . import reference  # noqa: F401
from .functional import (  # noqa: F401
    scatter_gather, indegree_norm, degree_scale, linear, relu, sigmoid,
    add, mul, dropout, softmax_cross_entropy, decode_metrics, adam_step,
    has_ext, set_dropout_seed, set_dropout_counter, reset_dropout_offset,
)
