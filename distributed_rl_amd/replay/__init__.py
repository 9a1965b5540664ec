from .per import (  # noqa: F401
    ReplayBase,
    TorchPER,
    FifoReplay,
    Schema,
    make_apex_schema,
    make_r2d2_schema,
)


class PER(TorchPER):
    """Name-compatible shim for the reference's ``baseline.PER.PER(maxlen,
    max_value, beta)`` surface: exposes push/sample/update/
    remove_to_fit/max_weight over the torch PER. The production GPU path is
    HipSumTreePER (gpu_per.py)."""

    def __init__(self, maxlen: int, max_value: float = 1.0, beta: float = 0.4,
                 schema=None, device: str = "cpu"):
        import torch

        super().__init__(maxlen, schema or {"blob": ((), torch.float32)}, device)
        self.beta = beta
        self.max_value = max_value

    @property
    def memory(self):
        return self.data

    @property
    def max_weight(self) -> float:
        n = len(self)
        p = self.priorities[self.priorities > 0]
        if n == 0 or p.numel() == 0:
            return 1.0
        min_prob = float(p.min()) / max(float(p.sum()), 1e-12)
        return (1.0 / (n * min_prob)) ** self.beta

    def remove_to_fit(self):
        """Ring storage evicts oldest-first on push; nothing to trim."""
        return None


def make_per(capacity, schema, device="cpu", **kw):
    """PER factory: HIP sum-tree on device, torch implementation on CPU."""
    import torch

    if str(device).startswith("cuda") and torch.cuda.is_available():
        from .gpu_per import HipSumTreePER

        return HipSumTreePER(capacity, schema, device, **kw)
    return TorchPER(capacity, schema, device)
