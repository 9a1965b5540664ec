from .base_agent import (  # noqa: F401
    BaseAgent,
    baseAgent,
    get_optim,
    getOptim,
    CNN2D,
    MLP,
    LSTMNET,
    ViewV2,
    Add,
    Mean,
    Substract,
)
