from . import native  # noqa: F401
from .functional import (  # noqa: F401
    conv2d,
    conv_pool_stem,
    dropout,
    linear,
    log_softmax,
    logsoftmax_nll,
    maxpool2x2_relu,
    relu,
)
from .sgd import SGD  # noqa: F401
