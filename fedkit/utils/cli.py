"""CLI override layer: every FedConfig field becomes a --flag.

The reference configures via module constants edited in-file (SURVEY.md §5
config system); here the same names are defaults overridable from the
command line so the script-level API stays recognizable.
"""

import argparse
import dataclasses

from ..parallel.runtime import FedConfig


def _parse_bool(v: str) -> bool:
    return str(v).lower() in ("1", "true", "yes", "on")


def config_from_cli(defaults: FedConfig, argv=None) -> FedConfig:
    ap = argparse.ArgumentParser()
    for f in dataclasses.fields(FedConfig):
        default = getattr(defaults, f.name)
        if f.type in ("bool", bool) or isinstance(default, bool):
            ap.add_argument(f"--{f.name}", type=_parse_bool, default=default)
        elif isinstance(default, int):
            ap.add_argument(f"--{f.name}", type=int, default=default)
        elif isinstance(default, float):
            ap.add_argument(f"--{f.name}", type=float, default=default)
        else:
            ap.add_argument(f"--{f.name}", type=str, default=default)
    args, _ = ap.parse_known_args(argv)
    return FedConfig(**{f.name: getattr(args, f.name)
                        for f in dataclasses.fields(FedConfig)})
