"""Config system tests (schema validation, defaults, lookup, validate)."""

import copy
import os

import pytest
import yaml

from msrflute_amd.config import ConfigValidationError, FLUTEConfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _load(name):
    with open(os.path.join(REPO, "configs", name)) as f:
        return yaml.safe_load(f)


@pytest.mark.parametrize("name", sorted(
    f for f in os.listdir(os.path.join(REPO, "configs")) if f.endswith(".yaml")))
def test_all_shipped_configs_validate(name):
    cfg = FLUTEConfig.from_dict(_load(name))
    assert cfg["model_config"]["model_type"]
    # defaults filled
    assert cfg["server_config"]["type"] in ("model_optimization", "personalization")
    assert isinstance(cfg["server_config"]["max_iteration"], int)


def test_missing_required_key_raises():
    cfg = _load("cv_lr_mnist.yaml")
    del cfg["server_config"]["optimizer_config"]
    with pytest.raises(ConfigValidationError):
        FLUTEConfig.from_dict(cfg)


def test_defaults_are_filled():
    cfg = _load("cv_lr_mnist.yaml")
    cfg["server_config"].pop("best_model_criterion", None)
    out = FLUTEConfig.from_dict(cfg)
    assert out["server_config"]["best_model_criterion"] == "loss"
    assert out["server_config"]["data_config"]["val"]["pin_memory"] is True


def test_forbidden_key_rejected():
    cfg = _load("cv_lr_mnist.yaml")
    cfg["server_config"]["data_config"]["num_clients"] = 5
    with pytest.raises(ConfigValidationError):
        FLUTEConfig.from_dict(cfg)


def test_lookup_and_attribute_access():
    cfg = FLUTEConfig.from_dict(_load("cv_lr_mnist.yaml"))
    assert cfg.lookup("server_config.optimizer_config.type") == "sgd"
    assert cfg.server_config.optimizer_config.type == "sgd"
    assert cfg.lookup("does.not.exist", 42) == 42


def test_validate_joins_paths():
    cfg = FLUTEConfig.from_dict(_load("cv_lr_mnist.yaml"))
    cfg["data_path"] = "/data"
    cfg["output_path"] = "/out"
    cfg["server_config"]["data_config"]["val"]["vocab_dict"] = "vocab.txt"
    cfg.validate()
    assert cfg["server_config"]["data_config"]["val"]["vocab_dict"] == \
        os.path.join("/data", "vocab.txt")


def test_num_clients_per_iteration_range_string():
    cfg = _load("cv_lr_mnist.yaml")
    cfg["server_config"]["num_clients_per_iteration"] = "5,10"
    out = FLUTEConfig.from_dict(cfg)
    assert out["server_config"]["num_clients_per_iteration"] == "5,10"


def test_to_dict_roundtrip():
    raw = _load("cv_lr_mnist.yaml")
    cfg = FLUTEConfig.from_dict(raw)
    d = cfg.to_dict()
    cfg2 = FLUTEConfig.from_dict(copy.deepcopy(d))
    assert cfg2.to_dict() == d
