"""Configuration system.

Loads the algorithm cfg JSON (schema-compatible with the reference's
``cfg/{ape_x,r2d2,impala}.json``; see /root/reference/configuration.py:36-110 for
the constants the reference exposes) WITHOUT import-time side effects:

* the reference selects the algorithm by editing a hardcoded path
  (``configuration.py:11-13``) and creates ``./log`` / ``./weight`` trees at import
  time (``configuration.py:16-32``). Here the algorithm is selected by CLI/env
  (``DRL_CFG`` or an explicit path) and directories are created lazily by the
  components that write to them.
"""

from __future__ import annotations

import json
import os
import time
from dataclasses import dataclass
from typing import Any, Dict, Optional

_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEFAULT_CFG_DIR = os.path.join(_REPO_ROOT, "cfg")

ALGORITHMS = ("APE_X", "R2D2", "IMPALA")


def load_json(path: str) -> Dict[str, Any]:
    with open(path, "r") as f:
        return json.load(f)


@dataclass
class Config:
    """Typed view over the reference cfg JSON schema.

    Every key of the reference schema is preserved verbatim in ``raw`` and
    surfaced as an attribute where the framework consumes it. Unknown keys are
    kept (schema is open, like ``from configuration import *``).
    """

    raw: Dict[str, Any]
    path: str = ""

    # ---- identity -----------------------------------------------------
    @property
    def alg(self) -> str:
        return str(self.raw["ALG"]).upper()

    # ---- core hyperparameters ----------------------------------------
    @property
    def action_size(self) -> int:
        return int(self.raw["ACTION_SIZE"])

    @property
    def gamma(self) -> float:
        return float(self.raw.get("GAMMA", 0.99))

    @property
    def batch_size(self) -> int:
        return int(self.raw.get("BATCHSIZE", 32))

    @property
    def unroll_step(self) -> int:
        return int(self.raw.get("UNROLL_STEP", 1))

    @property
    def replay_memory_len(self) -> int:
        return int(self.raw.get("REPLAY_MEMORY_LEN", 100000))

    @property
    def buffer_size(self) -> int:
        return int(self.raw.get("BUFFER_SIZE", 50000))

    # PER
    @property
    def use_per(self) -> bool:
        # reference: configuration.py:67  (use_per = ALG != "IMPALA")
        return self.alg != "IMPALA"

    @property
    def alpha(self) -> float:
        return float(self.raw.get("ALPHA", 0.6))

    @property
    def beta(self) -> float:
        return float(self.raw.get("BETA", 0.4))

    @property
    def target_frequency(self) -> int:
        return int(self.raw.get("TARGET_FREQUENCY", 2500))

    @property
    def num_actors(self) -> int:
        return int(self.raw.get("N", 8))

    # R2D2
    @property
    def fixed_trajectory(self) -> int:
        return int(self.raw.get("FIXED_TRAJECTORY", 80))

    @property
    def burn_in(self) -> int:
        return int(self.raw.get("MEM", 20))

    @property
    def use_rescaling(self) -> bool:
        return bool(self.raw.get("USE_RESCALING", False))

    # IMPALA
    @property
    def c_lambda(self) -> float:
        return float(self.raw.get("C_LAMBDA", 1.0))

    @property
    def c_value(self) -> float:
        return float(self.raw.get("C_VALUE", 1.0))

    @property
    def p_value(self) -> float:
        return float(self.raw.get("P_VALUE", 1.0))

    @property
    def entropy_r(self) -> float:
        return float(self.raw.get("ENTROPY_R", 0.01))

    @property
    def use_reward_clip(self) -> bool:
        return bool(self.raw.get("USE_REWARD_CLIP", False))

    # ---- devices ------------------------------------------------------
    @property
    def actor_device(self) -> str:
        return str(self.raw.get("DEVICE", "cpu"))

    @property
    def learner_device(self) -> str:
        return str(self.raw.get("LEARNER_DEVICE", "cuda:0"))

    # ---- nested sections ----------------------------------------------
    @property
    def optim_info(self) -> Dict[str, Any]:
        return dict(self.raw["optim"])

    @property
    def model_info(self) -> Dict[str, Any]:
        return dict(self.raw["model"])

    # ---- run bookkeeping ----------------------------------------------
    def run_name(self, now: Optional[time.struct_time] = None) -> str:
        """Timestamp-named run dir, same format the reference uses
        (configuration.py:101-102): MM_DD_YYYY_HH_MM_SS."""
        t = now or time.localtime()
        return time.strftime("%m_%d_%Y_%H_%M_%S", t)

    def log_dir(self, root: str = ".", run: Optional[str] = None) -> str:
        return os.path.join(root, "log", self.alg, run or self.run_name())

    def weight_dir(self, root: str = ".", run: Optional[str] = None) -> str:
        return os.path.join(root, "weight", self.alg, run or self.run_name())

    # ---- transport (replaces the reference's REDIS_SERVER keys) --------
    @property
    def transport_dir(self) -> str:
        """Rendezvous directory for the shared-memory transport.

        The reference's transport endpoints are the ``REDIS_SERVER*`` keys
        (cfg/ape_x.json:4-6). Our MI355X-native transport is shared-memory
        rings; the rendezvous is a filesystem path actors and the learner
        agree on (``DRL_TRANSPORT_DIR`` env overrides)."""
        return os.environ.get(
            "DRL_TRANSPORT_DIR", self.raw.get("TRANSPORT_DIR", "/dev/shm/drl")
        )

    def __getitem__(self, key: str) -> Any:
        return self.raw[key]

    def get(self, key: str, default: Any = None) -> Any:
        return self.raw.get(key, default)


def cfg_path_for(alg: str, cfg_dir: str = DEFAULT_CFG_DIR) -> str:
    name = f"{alg.lower()}.json"
    path = os.path.join(cfg_dir, name)
    if not os.path.exists(path):
        raise FileNotFoundError(f"no such cfg: {path}")
    return path


def load_config(spec: Optional[str] = None) -> Config:
    """Load a Config.

    ``spec`` may be an algorithm name ("ape_x"/"r2d2"/"impala"), a path to a
    cfg JSON, or None (falls back to the DRL_CFG env var, then ape_x).
    """
    if spec is None:
        spec = os.environ.get("DRL_CFG", "ape_x")
    if os.path.exists(spec) and spec.endswith(".json"):
        path = spec
    else:
        path = cfg_path_for(spec.replace("-", "_"))
    raw = load_json(path)
    return Config(raw=raw, path=path)
