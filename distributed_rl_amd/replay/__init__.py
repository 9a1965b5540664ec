from .per import (  # noqa: F401
    ReplayBase,
    TorchPER,
    FifoReplay,
    Schema,
    make_apex_schema,
    make_r2d2_schema,
)


def make_per(capacity, schema, device="cpu", **kw):
    """PER factory: HIP sum-tree on device, torch implementation on CPU."""
    import torch

    if str(device).startswith("cuda") and torch.cuda.is_available():
        from .gpu_per import HipSumTreePER

        return HipSumTreePER(capacity, schema, device, **kw)
    return TorchPER(capacity, schema, device)
