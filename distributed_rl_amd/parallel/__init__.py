from .ddp import FlatGradReducer, init_distributed, attach_reducer  # noqa: F401
