from . import cpu_ref, functional  # noqa: F401
from .functional import conv_pool, linear_act, softmax_xent, sgd_step  # noqa: F401
