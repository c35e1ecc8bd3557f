import os

import pytest
import yaml

from mine_amd.config import Config, default_config, load_config, merge_configs

CONFIG_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                          "configs")


def test_merge_order():
    default = {"a": 1, "b": 2, "c": 3}
    merged = merge_configs(default, {"b": 20}, {"c": 30})
    assert merged == {"a": 1, "b": 20, "c": 30}


def test_merge_rejects_unknown_key():
    with pytest.raises(KeyError):
        merge_configs({"a": 1}, {"zzz": 2})
    with pytest.raises(KeyError):
        merge_configs({"a": 1}, None, {"zzz": 2})


def test_load_dataset_configs():
    for name in ("llff", "realestate", "kitti_raw", "flowers", "dtu"):
        cfg = load_config(os.path.join(CONFIG_DIR, f"params_{name}.yaml"))
        assert "data.img_h" in cfg
        assert isinstance(cfg["lr.decay_steps"], list)
        assert all(isinstance(x, int) for x in cfg["lr.decay_steps"])


def test_extra_config_json():
    cfg = load_config(os.path.join(CONFIG_DIR, "params_llff.yaml"),
                      extra_config='{"data.img_h": 64}')
    assert cfg["data.img_h"] == 64


def test_config_immutable_and_replace():
    cfg = default_config()
    with pytest.raises(TypeError):
        cfg["data.img_h"] = 1  # Mapping is read-only
    cfg2 = cfg.replace(**{"data.img_h": 64})
    assert cfg2["data.img_h"] == 64
    assert cfg["data.img_h"] != 64 or cfg["data.img_h"] == 384


def test_dump_yaml_roundtrip(tmp_path):
    cfg = default_config()
    p = tmp_path / "params.yaml"
    cfg.dump_yaml(str(p))
    with open(p) as f:
        loaded = yaml.safe_load(f)
    assert loaded["data.img_h"] == cfg["data.img_h"]
    assert Config(loaded)["mpi.num_bins_coarse"] == cfg["mpi.num_bins_coarse"]
