"""CLI command framework: subclasses of Command auto-register by name.

Same user surface as the reference's metaclass-registered commands
(/root/reference/distllm/cli_api/base.py:1-29), realized with
__init_subclass__ instead of a metaclass.
"""
from __future__ import annotations

import argparse
from typing import Dict, Type

commands: Dict[str, Type["Command"]] = {}


class Command:
    name: str = ""
    help: str = ""

    def __init_subclass__(cls, **kw):
        super().__init_subclass__(**kw)
        if cls.name:
            commands[cls.name] = cls

    def configure(self, parser: argparse.ArgumentParser) -> None:
        """Add command-specific arguments."""

    def __call__(self, args: argparse.Namespace) -> int:
        raise NotImplementedError
