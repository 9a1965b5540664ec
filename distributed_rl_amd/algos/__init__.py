from typing import Dict, Tuple

from ..config import Config


def get_learner_cls(alg: str):
    alg = alg.upper()
    if alg == "APE_X":
        from .ape_x import ApexLearner

        return ApexLearner
    if alg == "R2D2":
        from .r2d2 import R2D2Learner

        return R2D2Learner
    if alg == "IMPALA":
        from .impala import ImpalaLearner

        return ImpalaLearner
    raise ValueError(f"unknown algorithm {alg}")


def get_player_cls(alg: str):
    alg = alg.upper()
    if alg == "APE_X":
        from .ape_x import ApexPlayer

        return ApexPlayer
    if alg == "R2D2":
        from .r2d2 import R2D2Player

        return R2D2Player
    if alg == "IMPALA":
        from .impala import ImpalaPlayer

        return ImpalaPlayer
    raise ValueError(f"unknown algorithm {alg}")


def get_vec_runner(alg: str):
    """Vectorized multi-env actor loop for `alg`, or None if the algorithm
    only supports one env per actor process (fleet envs_per_proc > 1
    requires this)."""
    if alg.upper() == "APE_X":
        from .ape_x import run_apex_vec

        return run_apex_vec
    return None


def get_wire_schema(cfg: Config) -> Tuple[Dict, bool]:
    """(schema, with_priority) for the transport record codec."""
    alg = cfg.alg
    if alg == "APE_X":
        from ..replay import make_apex_schema

        return make_apex_schema(), True
    if alg == "R2D2":
        from ..replay import make_r2d2_schema

        hidden = 512
        for node in cfg.model_info.values():
            if str(node.get("netCat", "")).upper() == "LSTMNET":
                hidden = int(node["hiddenSize"])
        return make_r2d2_schema(cfg.fixed_trajectory, hidden=hidden), True
    if alg == "IMPALA":
        from .impala import make_impala_schema

        return make_impala_schema(cfg.unroll_step), False
    raise ValueError(alg)
