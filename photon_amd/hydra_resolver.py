"""Resolve the config tree and dump ``config.yaml`` for every other process.

Parity with reference ``photon/hydra_resolver.py:15-39``: the launch script
runs ``python -m photon_amd.hydra_resolver <overrides...>`` once; the resolved
YAML lands in ``$PHOTON_SAVE_PATH/config.yaml`` and is the single source of
truth every process (server, clients, bench) re-loads.
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

from .conf import compose, config_yaml_dir, dump, validate


def main(argv: list[str] | None = None) -> Path:
    overrides = list(sys.argv[1:] if argv is None else argv)
    cfg = validate(compose(config_yaml_dir(), "base", overrides))
    save_path = Path(os.environ.get("PHOTON_SAVE_PATH", "."))
    out = save_path / "config.yaml"
    dump(cfg, out)
    print(f"wrote {out}")
    return out


if __name__ == "__main__":
    main()
