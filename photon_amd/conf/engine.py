"""Hydra-compatible config composition engine.

The reference resolves its config with Hydra 1.3 + OmegaConf
(photon/hydra_resolver.py:15-39) and every process re-loads the dumped
``config.yaml``.  Neither hydra nor omegaconf ships in this image, so this
module implements the subset of the surface the reference uses:

* a YAML tree with a ``defaults`` list in the primary config
  (``- _self_``, ``- group: option``, ``- group@target.path: option``),
* ``${path.to.key}`` interpolation (absolute paths, resolved on the merged
  tree, recursively),
* CLI overrides: ``key.path=value`` (set), ``+key=value`` (add),
  ``++key=value`` (force add), ``~key`` / ``~key=value`` (delete),
  ``group=option`` / ``group@target: option`` (defaults-list override),
* dump of the fully-resolved tree to ``config.yaml``.

Values in overrides are parsed as YAML scalars (so ``lr=1e-4`` is a float,
``flag=true`` a bool, ``layers=[a,b]`` a list), matching Hydra behavior.
"""

from __future__ import annotations

import copy
import re
from pathlib import Path
from typing import Any, Iterator

import yaml


class ConfigError(Exception):
    pass


_MISSING = object()


class DictConfig(dict):
    """A dict with attribute access and dotted-path helpers (OmegaConf-like)."""

    def __getattr__(self, name: str) -> Any:
        try:
            return self[name]
        except KeyError as e:  # pragma: no cover - defensive
            raise AttributeError(name) from e

    def __setattr__(self, name: str, value: Any) -> None:
        self[name] = value

    def __delattr__(self, name: str) -> None:
        try:
            del self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    # -- dotted path access -------------------------------------------------
    def select(self, path: str, default: Any = _MISSING) -> Any:
        node: Any = self
        for part in path.split("."):
            if isinstance(node, dict) and part in node:
                node = node[part]
            else:
                if default is _MISSING:
                    raise ConfigError(f"missing config key: {path!r}")
                return default
        return node

    def set_path(self, path: str, value: Any, *, create: bool = True) -> None:
        parts = path.split(".")
        node: Any = self
        for part in parts[:-1]:
            if not isinstance(node, dict):
                raise ConfigError(f"cannot descend into non-dict at {part!r} of {path!r}")
            if part not in node or not isinstance(node[part], dict):
                if not create:
                    raise ConfigError(f"missing parent {part!r} for {path!r}")
                node[part] = DictConfig()
            node = node[part]
        node[parts[-1]] = value

    def del_path(self, path: str) -> None:
        parts = path.split(".")
        node: Any = self
        for part in parts[:-1]:
            if not (isinstance(node, dict) and part in node):
                raise ConfigError(f"cannot delete missing key {path!r}")
            node = node[part]
        if not (isinstance(node, dict) and parts[-1] in node):
            raise ConfigError(f"cannot delete missing key {path!r}")
        del node[parts[-1]]

    def to_plain(self) -> dict:
        return _to_plain(self)


def _to_plain(node: Any) -> Any:
    if isinstance(node, dict):
        return {k: _to_plain(v) for k, v in node.items()}
    if isinstance(node, list):
        return [_to_plain(v) for v in node]
    return node


def _wrap(node: Any) -> Any:
    if isinstance(node, DictConfig):
        for k, v in node.items():
            node[k] = _wrap(v)
        return node
    if isinstance(node, dict):
        return DictConfig({k: _wrap(v) for k, v in node.items()})
    if isinstance(node, list):
        return [_wrap(v) for v in node]
    return node


def _merge(dst: DictConfig, src: dict) -> DictConfig:
    """Deep-merge src into dst (src wins; dicts merge recursively)."""
    for k, v in src.items():
        if isinstance(v, dict) and isinstance(dst.get(k), dict):
            _merge(dst[k], v)
        else:
            dst[k] = _wrap(copy.deepcopy(v))
    return dst


def _load_yaml(path: Path, *, allow_list: bool = False) -> Any:
    with open(path) as f:
        data = yaml.safe_load(f)
    if data is None:
        data = {}
    if not isinstance(data, dict) and not (allow_list and isinstance(data, list)):
        raise ConfigError(f"{path} must contain a mapping, got {type(data)}")
    return data


def _parse_value(text: str) -> Any:
    try:
        value = yaml.safe_load(text)
    except yaml.YAMLError:
        return text
    if isinstance(value, str):
        # YAML 1.1 misses bare scientific notation ("1e-5"); Hydra treats it
        # as a float — match that.
        try:
            return int(value)
        except ValueError:
            pass
        try:
            return float(value)
        except ValueError:
            pass
    return value


_INTERP_RE = re.compile(r"\$\{([A-Za-z0-9_.@-]+)\}")


def _resolve_node(node: Any, root: DictConfig, stack: tuple[str, ...]) -> Any:
    if isinstance(node, dict):
        for k in list(node.keys()):
            node[k] = _resolve_node(node[k], root, stack)
        return node
    if isinstance(node, list):
        return [_resolve_node(v, root, stack) for v in node]
    if isinstance(node, str):
        m = _INTERP_RE.fullmatch(node)
        if m:  # whole-string interpolation preserves type
            return _resolve_ref(m.group(1), root, stack)

        def sub(match: re.Match) -> str:
            return str(_resolve_ref(match.group(1), root, stack))

        return _INTERP_RE.sub(sub, node)
    return node


def _resolve_ref(path: str, root: DictConfig, stack: tuple[str, ...]) -> Any:
    if path in stack:
        raise ConfigError(f"interpolation cycle: {' -> '.join(stack + (path,))}")
    value = root.select(path)
    return _resolve_node(copy.deepcopy(value), root, stack + (path,))


def resolve(cfg: DictConfig) -> DictConfig:
    """Resolve all ``${...}`` interpolations in-place and return cfg."""
    _resolve_node(cfg, cfg, ())
    return cfg


class _Default:
    __slots__ = ("group", "option", "target", "is_self")

    def __init__(self, group: str | None, option: str | None, target: str | None, is_self: bool):
        self.group, self.option, self.target, self.is_self = group, option, target, is_self


def _parse_defaults(defaults: list) -> list[_Default]:
    out: list[_Default] = []
    for entry in defaults:
        if entry == "_self_":
            out.append(_Default(None, None, None, True))
            continue
        if not isinstance(entry, dict) or len(entry) != 1:
            raise ConfigError(f"bad defaults entry: {entry!r}")
        (key, option), = entry.items()
        if "@" in key:
            group, target = key.split("@", 1)
        else:
            group, target = key, key.replace("/", ".")
        out.append(_Default(group, option, target, False))
    return out


def compose(
    config_dir: str | Path,
    config_name: str = "base",
    overrides: list[str] | None = None,
) -> DictConfig:
    """Compose the config tree like ``hydra.compose`` and resolve interpolations."""
    config_dir = Path(config_dir)
    overrides = list(overrides or [])

    primary = _load_yaml(config_dir / f"{config_name}.yaml")
    defaults = _parse_defaults(primary.pop("defaults", ["_self_"]))
    if not any(d.is_self for d in defaults):
        defaults.insert(0, _Default(None, None, None, True))

    # Partition overrides: group overrides (match a defaults-list group),
    # deletes (~), adds (+/++), plain sets.
    group_names = {d.group for d in defaults if not d.is_self}
    sets: list[tuple[str, Any]] = []
    adds: list[tuple[str, Any]] = []
    deletes: list[str] = []
    for ov in overrides:
        if ov.startswith("~"):
            deletes.append(ov[1:].split("=", 1)[0])
            continue
        force = ov.startswith("++")
        add = ov.startswith("+") and not force
        body = ov.lstrip("+")
        if "=" not in body:
            raise ConfigError(f"override {ov!r} has no '='")
        key, raw = body.split("=", 1)
        value = _parse_value(raw)
        base_group = key.split("@", 1)[0]
        if base_group in group_names and isinstance(value, str):
            # defaults-list override: llm_config=mpt-1b
            for d in defaults:
                if d.group == base_group and (
                    "@" not in key or key.split("@", 1)[1] == d.target
                ):
                    d.option = value
                    break
            continue
        if add or force:
            adds.append((key, value))
        else:
            sets.append((key, value))

    cfg = DictConfig()
    for d in defaults:
        if d.is_self:
            _merge(cfg, primary)
            continue
        path = config_dir / d.group / f"{d.option}.yaml"
        if not path.exists():
            raise ConfigError(f"missing config group file: {path}")
        sub = _load_yaml(path, allow_list=True)
        if isinstance(sub, list):
            # list-valued group file (e.g. dataset/streams/8_clients.yaml):
            # set verbatim at the target path.
            cfg.set_path(d.target, _wrap(copy.deepcopy(sub)))
        elif d.target in (None, "", "_global_"):
            _merge(cfg, sub)
        else:
            node = DictConfig()
            _merge(node, sub)
            existing = cfg.select(d.target, default=None)
            if isinstance(existing, dict):
                _merge(existing, node)
            else:
                cfg.set_path(d.target, node)

    for key, value in sets:
        cfg.set_path(key, _wrap(value))
    for key, value in adds:
        cfg.set_path(key, _wrap(value))
    for key in deletes:
        cfg.del_path(key)

    return resolve(cfg)


def load_resolved(path: str | Path) -> DictConfig:
    """Load an already-resolved config.yaml (the per-process reload pattern,
    photon/server_app.py:116)."""
    cfg = _wrap(_load_yaml(Path(path)))
    return cfg


def dump(cfg: DictConfig, path: str | Path) -> None:
    """Dump the resolved tree to YAML (photon/hydra_resolver.py:30-39 pattern)."""
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    with open(path, "w") as f:
        yaml.safe_dump(cfg.to_plain(), f, default_flow_style=False, sort_keys=False)


def config_yaml_dir() -> Path:
    return Path(__file__).parent / "yaml"


def iter_leaves(cfg: DictConfig, prefix: str = "") -> Iterator[tuple[str, Any]]:
    for k, v in cfg.items():
        p = f"{prefix}.{k}" if prefix else str(k)
        if isinstance(v, dict):
            yield from iter_leaves(v, p)
        else:
            yield p, v
