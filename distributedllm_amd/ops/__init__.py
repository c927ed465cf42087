"""Native engine extension loader.

Loads the in-tree `_core.so` (built by `build.py` / `__graft_entry__.build`).
On a machine WITH a GPU, a missing or unloadable extension is a hard error —
there is deliberately no silent eager/PyTorch fallback on the GPU path, so
GPU tests can never pass on anything but the HIP kernels. On CPU-only
machines `core()` raises and callers use the torch reference engine instead.
"""
from __future__ import annotations

import importlib.machinery
import importlib.util
from pathlib import Path

_SO = Path(__file__).resolve().parent / "_core.so"
_mod = None
_err: Exception | None = None


def _load():
    global _mod, _err
    if _mod is not None or _err is not None:
        return
    try:
        import torch  # noqa: F401  (extension links against torch libs)
        if not _SO.exists():
            raise FileNotFoundError(
                f"native engine extension not built: {_SO} missing — run "
                "`python -m distributedllm_amd.ops.build` (or __graft_entry__"
                ".build())")
        loader = importlib.machinery.ExtensionFileLoader("_core", str(_SO))
        spec = importlib.util.spec_from_loader("_core", loader)
        mod = importlib.util.module_from_spec(spec)
        loader.exec_module(mod)
        _mod = mod
    except Exception as e:  # noqa: BLE001
        _err = e


def available() -> bool:
    _load()
    return _mod is not None


def core():
    """The native extension module; raises loudly if it cannot be loaded."""
    _load()
    if _mod is None:
        raise RuntimeError(
            f"distributedllm_amd native engine unavailable: {_err}") from _err
    return _mod
