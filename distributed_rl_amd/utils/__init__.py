"""Utility surface matching the reference's ``baseline.utils`` API
(call sites: /root/reference/configuration.py:36-37,
IMPALA/Learner.py:11, IMPALA/ReplayMemory.py:21)."""

from __future__ import annotations

import json
import logging
import pickle
import random
from collections import deque
from typing import Any, Dict, List, Sequence


def dumps(obj: Any) -> bytes:
    return pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)


def loads(blob: bytes) -> Any:
    return pickle.loads(blob)


class jsonParser:
    """``jsonParser(path).loadParser() -> dict`` (configuration.py:36-37)."""

    def __init__(self, path: str):
        self.path = path

    def loadParser(self) -> Dict[str, Any]:
        with open(self.path, "r") as f:
            return json.load(f)


class writeTrainInfo:
    """Pretty-printable config block for TensorBoard text
    (APE_X/Learner.py:36-39 uses ``.info``)."""

    def __init__(self, cfg: Dict[str, Any]):
        lines = ["Configuration:", ""]
        for k, v in cfg.items():
            lines.append(f"    {k}: {v}")
        self.info = "\n".join(lines)

    def __str__(self) -> str:
        return self.info


def setup_logger(name: str = "drl", level: int = logging.INFO) -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        h = logging.StreamHandler()
        h.setFormatter(
            logging.Formatter("%(asctime)s %(name)s %(levelname)s %(message)s")
        )
        logger.addHandler(h)
    logger.setLevel(level)
    return logger


class ReplayMemory:
    """Uniform FIFO replay (``baseline.utils.ReplayMemory``; consumed at
    IMPALA/ReplayMemory.py:21,32,68,85): push(list), sample(n), len."""

    def __init__(self, maxlen: int):
        self.memory: deque = deque(maxlen=int(maxlen))

    def push(self, items: Sequence[Any]) -> None:
        self.memory.extend(items)

    def sample(self, n: int) -> List[Any]:
        return random.sample(self.memory, n)

    def __len__(self) -> int:
        return len(self.memory)
