from .layers import (  # noqa: F401
    GELU,
    MLP,
    LayerNorm,
    Linear,
    MSELoss,
    Module,
    Parameter,
    ReLU,
    Sequential,
    Softmax,
    SoftmaxMSE,
    SoftmaxXent,
)
from .optimizer import SGD, AdamW  # noqa: F401
