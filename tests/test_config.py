"""Config engine tests — Hydra-surface parity."""

import pytest

from photon_amd.conf import (
    ConfigError,
    compose,
    config_yaml_dir,
    dump,
    load_resolved,
    validate,
)


def test_base_composition():
    cfg = validate(compose(config_yaml_dir(), "base", []))
    assert cfg.fl.strategy_name == "NESTOROV"
    assert cfg.llm_config.model.d_model == 768
    assert len(cfg.dataset.train.streams) == 8
    assert cfg.llm_config.model.max_seq_len == 2048  # ${} interpolation
    assert cfg.wandb.setup.name == cfg.run_uuid


def test_group_override_and_deletes():
    cfg = compose(
        config_yaml_dir(),
        "base",
        [
            "llm_config=mpt-1b",
            "fl.n_rounds=5",
            "~llm_config.fsdp_config",
            "dataset/streams@dataset.train.streams=4_clients",
            "++llm_config.custom_flag=7",
        ],
    )
    assert cfg.llm_config.model.d_model == 2048
    assert cfg.llm_config.model.n_heads == 16  # d_head == 128
    assert cfg.fl.n_rounds == 5
    assert "fsdp_config" not in cfg.llm_config
    assert len(cfg.dataset.train.streams) == 4
    assert cfg.llm_config.custom_flag == 7


def test_override_value_types():
    cfg = compose(
        config_yaml_dir(),
        "base",
        ["llm_config.optimizer.lr=1e-5", "use_wandb=true", "fl.random_layers=[a,b]"],
    )
    assert cfg.llm_config.optimizer.lr == 1e-5
    assert cfg.use_wandb is True
    assert cfg.fl.random_layers == ["a", "b"]


def test_dump_and_reload(tmp_path):
    cfg = compose(config_yaml_dir(), "base", ["run_uuid=dumptest"])
    out = tmp_path / "config.yaml"
    dump(cfg, out)
    cfg2 = load_resolved(out)
    assert cfg2.run_uuid == "dumptest"
    assert cfg2.to_plain() == cfg.to_plain()


def test_all_model_presets_compose():
    for preset in ("mpt-125m", "mpt-350m", "mpt-1b", "mpt-3b", "mpt-7b"):
        cfg = validate(compose(config_yaml_dir(), "base", [f"llm_config={preset}"]))
        m = cfg.llm_config.model
        assert m.vocab_size == 50368
        assert m.d_model % m.n_heads == 0


def test_validate_rejects_bad_strategy():
    with pytest.raises(ConfigError):
        validate(compose(config_yaml_dir(), "base", ["fl.strategy_name=BOGUS"]))


def test_validate_rejects_legacy_comm_stack():
    with pytest.raises(ConfigError):
        validate(compose(config_yaml_dir(), "base", ["photon.comm_stack.shm=true"]))


def test_resolver_cli(tmp_path, monkeypatch):
    monkeypatch.setenv("PHOTON_SAVE_PATH", str(tmp_path))
    from photon_amd.hydra_resolver import main

    out = main(["fl.n_rounds=3"])
    cfg = load_resolved(out)
    assert cfg.fl.n_rounds == 3


# -- property fuzzing --------------------------------------------------------
from hypothesis import given, settings
from hypothesis import strategies as hst

from photon_amd.conf.schema import duration_to_batches


@settings(max_examples=30, deadline=None)
@given(n=hst.integers(0, 10**7))
def test_duration_parse_property(n):
    assert duration_to_batches(f"{n}ba") == n
    assert duration_to_batches(n) == n
    assert duration_to_batches(str(n)) == n


def test_v03_presets_compose():
    """The vendored reference tasks_v0.3 / eval_gauntlet_v0.3 groups load
    through the hydra-surface engine (the reference's exact preset names)."""
    from photon_amd.conf import compose, config_yaml_dir

    cfg = compose(config_yaml_dir(), "base",
                  ["icl_tasks_config=tasks_v0.3",
                   "eval_gauntlet_config=eval_gauntlet_v0.3"])
    tasks = cfg["icl_tasks_config"]["icl_tasks"]
    assert len(tasks) == 32
    kinds = {str(t["icl_task_type"]) for t in tasks}
    assert kinds == {"language_modeling", "multiple_choice", "schema",
                     "generation_task_with_answers"}
    g = cfg["eval_gauntlet_config"]["eval_gauntlet"]
    assert len(g["categories"]) == 5
    assert "core_average" in g["averages"]
