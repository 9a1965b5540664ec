import json
import os

from distributed_rl_amd.config import load_config, cfg_path_for


def test_load_all_algorithms():
    for alg, nexp in (("ape_x", 6), ("r2d2", 6), ("impala", 7)):
        cfg = load_config(alg)
        assert cfg.action_size == 6
        assert cfg.batch_size == 32
        assert "model" in cfg.raw
        out_nodes = [k for k, v in cfg.model_info.items() if v.get("output")]
        assert len(out_nodes) == 1


def test_reference_schema_loads_unmodified():
    """The reference's own cfg files must load through our Config (schema
    byte-compat contract, BASELINE.json north_star)."""
    ref_cfg = "/root/reference/cfg"
    if not os.path.isdir(ref_cfg):
        return
    for name in ("ape_x.json", "r2d2.json", "impala.json"):
        cfg = load_config(os.path.join(ref_cfg, name))
        assert cfg.alg in ("APE_X", "R2D2", "IMPALA")
        assert cfg.action_size == 6
        assert cfg.optim_info["name"] in ("rmsprop", "adam")


def test_per_flags():
    assert load_config("ape_x").use_per
    assert load_config("r2d2").use_per
    assert not load_config("impala").use_per


def test_explicit_path(tmp_path):
    src = json.load(open(cfg_path_for("APE_X")))
    src["BATCHSIZE"] = 64
    p = tmp_path / "custom.json"
    p.write_text(json.dumps(src))
    cfg = load_config(str(p))
    assert cfg.batch_size == 64


def test_r2d2_keys():
    cfg = load_config("r2d2")
    assert cfg.fixed_trajectory == 80
    assert cfg.burn_in == 20
    assert cfg.use_rescaling
    assert cfg.gamma == 0.997
