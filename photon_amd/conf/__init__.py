from .engine import (
    ConfigError,
    DictConfig,
    compose,
    config_yaml_dir,
    dump,
    load_resolved,
    resolve,
)
from .schema import duration_to_batches, validate

__all__ = [
    "ConfigError",
    "DictConfig",
    "compose",
    "config_yaml_dir",
    "dump",
    "load_resolved",
    "resolve",
    "validate",
    "duration_to_batches",
]
