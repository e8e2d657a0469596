"""Config system tests (parity: reference tests/test_configs.py)."""

import pytest
import yaml

from trlx_amd import TRLConfig
from trlx_amd.data.default_configs import (
    default_ilql_config,
    default_ppo_config,
    default_sft_config,
)


@pytest.mark.parametrize("factory", [default_ppo_config, default_ilql_config, default_sft_config])
def test_defaults_roundtrip(factory):
    config = factory()
    d = config.to_dict()
    restored = TRLConfig.from_dict(d)
    assert restored.to_dict() == d


def test_yaml_roundtrip(tmp_path):
    config = default_ppo_config()
    p = tmp_path / "config.yml"
    with open(p, "w") as f:
        yaml.safe_dump(config.to_dict(), f)
    loaded = TRLConfig.load_yaml(str(p))
    assert loaded.method.ppo_epochs == config.method.ppo_epochs
    assert loaded.train.seq_length == config.train.seq_length


def test_update_dot_paths():
    config = default_ppo_config()
    updated = TRLConfig.update(config.to_dict(), {"method.ppo_epochs": 9, "train.batch_size": 7})
    assert updated.method.ppo_epochs == 9
    assert updated.train.batch_size == 7
    with pytest.raises(ValueError):
        TRLConfig.update(config.to_dict(), {"nonexistent.not_a_key": 1})


def test_evolve():
    config = default_ppo_config()
    evolved = config.evolve(train=dict(seq_length=512))
    assert evolved.train.seq_length == 512
    assert config.train.seq_length != 512 or config.train.seq_length == 1024


def test_canonical_ppo_hyperparameters():
    """The algorithmic parity anchor: reference defaults."""
    m = default_ppo_config().method
    assert m.num_rollouts == 128
    assert m.chunk_size == 128
    assert m.ppo_epochs == 4
    assert m.init_kl_coef == 0.001
    assert m.gamma == 1.0
    assert m.lam == 0.95
    assert m.cliprange == 0.2


def test_canonical_ilql_hyperparameters():
    m = default_ilql_config().method
    assert m.tau == 0.7
    assert m.gamma == 0.99
    assert m.cql_scale == 0.1
    assert m.awac_scale == 1
    assert m.alpha == 0.001
    assert m.steps_for_target_q_sync == 5
    assert m.two_qs


def test_all_shipped_configs_load():
    """Every YAML under configs/ loads into TRLConfig
    (parity: reference tests/test_configs.py over configs/**)."""
    import glob
    import os

    root = os.path.join(os.path.dirname(__file__), "..", "configs")
    paths = sorted(glob.glob(os.path.join(root, "*.yml")))
    assert len(paths) >= 6
    for p in paths:
        cfg = TRLConfig.load_yaml(p)
        assert cfg.train.seq_length > 0
