import os

import pytest

from yet_another_mobilenet_series_amd.utils.config import (
    AttrDict, default_config, setup_config)


def test_attrdict_nested_access():
    d = AttrDict({"a": {"b": 1}, "c": [1, {"d": 2}]})
    assert d.a.b == 1
    assert d.c[1].d == 2
    d.a.b = 5
    assert d["a"]["b"] == 5


def test_defaults_complete():
    cfg = default_config()
    assert cfg.model.name
    assert cfg.optim.lr > 0
    assert cfg.train.epochs > 0


def test_setup_config_app_and_overrides(tmp_path):
    p = tmp_path / "app.yml"
    p.write_text("model: {name: mobilenet_v1}\noptim: {lr: 0.5}\n")
    cfg = setup_config([f"app:{p}", "optim.lr=0.25", "train.epochs=3"])
    assert cfg.model.name == "mobilenet_v1"
    assert cfg.optim.lr == 0.25
    assert cfg.train.epochs == 3


def test_override_value_types():
    cfg = setup_config(["optim.nesterov=false", "optim.lr=1e-2",
                        "optim.multistep_milestones=[1,2]"])
    assert cfg.optim.nesterov is False
    assert cfg.optim.lr == 0.01
    assert cfg.optim.multistep_milestones == [1, 2]


def test_unknown_key_rejected():
    with pytest.raises(KeyError):
        setup_config(["optim.typo_key=1"])


def test_new_key_with_plus():
    cfg = setup_config(["+custom.flag=7"])
    assert cfg.custom.flag == 7


def test_shipped_app_configs_parse():
    apps = os.path.join(os.path.dirname(__file__), "..", "apps")
    for fn in sorted(os.listdir(apps)):
        if fn.endswith(".yml"):
            cfg = setup_config([f"app:{os.path.join(apps, fn)}"])
            assert cfg.model.name
