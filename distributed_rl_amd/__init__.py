"""distributed_rl_amd — MI355X-native distributed actor-learner RL engine.

A from-scratch framework with the capabilities of seungju-k1m/Distributed_RL
(Ape-X DQN, R2D2, IMPALA on Atari-shaped observations), re-architected for
AMD Instinct MI355X (gfx950 / CDNA4):

* GPU-resident lock-free prioritized replay (sum-tree in HBM3E) instead of a
  Redis + host-thread replay pipeline,
* hand-written HIP kernels for the learner hot path (dequant, fused TD loss,
  V-trace scan, value rescaling, sequence priorities, grad clip),
* CPU actor fleet feeding the learner over shared-memory rings + pinned
  hipMemcpyAsync staging instead of pickled Redis lists,
* learner data-parallelism over torch.distributed (RCCL over xGMI).
"""

__version__ = "0.1.0"

from .config import Config, load_config  # noqa: F401
