"""The reference ``baseline.utils`` parity surface:
jsonParser, writeTrainInfo, pickle helpers, ReplayMemory, setup_logger."""

import json

from distributed_rl_amd import utils
from distributed_rl_amd.config import DEFAULT_CFG_DIR, cfg_path_for


def test_json_parser_loads_cfg():
    path = cfg_path_for("ape_x", DEFAULT_CFG_DIR)
    d = utils.jsonParser(path).loadParser()
    assert d == json.load(open(path))
    assert d["ALG"] == "APE_X"


def test_write_train_info_renders_every_key():
    cfg = {"ALG": "APE_X", "GAMMA": 0.997, "nested": {"a": 1}}
    info = utils.writeTrainInfo(cfg)
    assert str(info) == info.info
    for k in cfg:
        assert k in info.info


def test_pickle_helpers_roundtrip():
    obj = {"x": [1, 2, 3], "y": "z"}
    assert utils.loads(utils.dumps(obj)) == obj


def test_replay_memory_fifo_eviction_and_sample():
    rm = utils.ReplayMemory(4)
    rm.push([1, 2, 3, 4, 5])
    assert len(rm) == 4 and list(rm.memory) == [2, 3, 4, 5]
    s = rm.sample(2)
    assert len(s) == 2 and all(v in (2, 3, 4, 5) for v in s)


def test_setup_logger_idempotent():
    a = utils.setup_logger("drl-test")
    b = utils.setup_logger("drl-test")
    assert a is b and len(a.handlers) == 1
