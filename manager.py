#!/usr/bin/env python3
"""CLI entry point (reference parity: /root/reference/manager.py:1-5)."""
import sys

from distributedllm_amd.cli import execute_command

if __name__ == "__main__":
    sys.exit(execute_command())
