"""CLI entry: `python manager.py <command> ...`.

Parity with the reference's parse_all_args/execute_command
(/root/reference/distllm/cli_api/__init__.py:9-24).
"""
from __future__ import annotations

import argparse
from typing import List, Optional

# note: import the submodule BEFORE the registry dict of the same name —
# the dict would otherwise shadow the submodule for `from . import commands`
from . import commands as _commands  # noqa: F401  (registers all commands)
from .base import commands


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="manager.py",
        description="distributedllm_amd — MI355X-native layer-sliced "
                    "LLM inference")
    sub = parser.add_subparsers(dest="command", required=True)
    for name, cls in sorted(commands.items()):
        cmd = cls()
        p = sub.add_parser(name, help=cls.help)
        cmd.configure(p)
        p.set_defaults(_cmd=cmd)
    return parser


def _operational_errors():
    from ..cluster.client import OperationFailedError  # noqa: F401
    from ..cluster.uploads import UploadError  # noqa: F401
    return (ConnectionError, OSError, ValueError, KeyError,
            OperationFailedError, UploadError)


def execute_command(argv: Optional[List[str]] = None) -> int:
    args = build_parser().parse_args(argv)
    try:
        return args._cmd(args)
    except KeyboardInterrupt:
        return 130
    except _operational_errors() as e:
        # operational failures (node down, bad config, malformed file)
        # get one clear line, not a traceback (reference CLI behavior:
        # cli_api/__init__.py prints argparse errors but lets everything
        # else crash)
        import sys as _sys
        print(f"error: {type(e).__name__}: {e}", file=_sys.stderr)
        return 1
