"""Declarative extra-CLI-flag spec (reference AddonParser,
tools/train_utils.py:14-44): a model module can declare additional
hyperparameter flags as data; the driver merges them into argparse."""
from __future__ import annotations

import argparse
from dataclasses import dataclass
from typing import Any, Optional, Sequence


@dataclass
class Addon:
    field: str
    default: Any = None
    type: Optional[type] = None
    action: Optional[str] = None
    help: str = ""


class AddonParser:
    def __init__(self, addons: Sequence[Addon]):
        self.addons = list(addons)

    def append(self, parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
        for a in self.addons:
            kwargs = {"default": a.default, "help": a.help}
            if a.action:
                kwargs["action"] = a.action
            elif a.type:
                kwargs["type"] = a.type
            elif a.default is not None:
                kwargs["type"] = type(a.default)
            parser.add_argument(f"--{a.field}", **kwargs)
        return parser

    def extract(self, args: argparse.Namespace) -> dict:
        return {a.field: getattr(args, a.field) for a in self.addons}
