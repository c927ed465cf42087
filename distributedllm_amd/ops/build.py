"""In-tree build of the native engine extension (`_core.so`).

Drives hipcc directly (no JIT cache under ~/.cache) so the built .so lives
in-tree and travels with the repo snapshot to GPU boxes. Cross-compiles for
gfx950 on machines without a GPU.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
SO_PATH = PKG_DIR / "_core.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

SOURCES = [CSRC / "kernels.hip", CSRC / "engine_ext.cpp"]
HEADERS = [CSRC / "kernels.h"]


def _torch_paths():
    import torch
    import torch.utils.cpp_extension as ce
    inc = ce.include_paths(device_type="cuda")
    lib = ce.library_paths(device_type="cuda")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return inc, lib, abi


def needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    return any(p.stat().st_mtime > so_mtime for p in SOURCES + HEADERS)


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_rebuild():
        return SO_PATH
    inc, lib, abi = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    hipcc = os.environ.get("HIPCC", "hipcc")

    cflags = [
        "-O3", "-std=c++17", "-fPIC",
        f"--offload-arch={ARCH}",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_core",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fno-gpu-rdc",
        "-Wno-ignored-attributes",
    ]
    incflags = [f"-I{p}" for p in inc + [str(CSRC), py_inc]]

    objs = []
    for src in SOURCES:
        obj = CSRC / (src.stem + ".o")
        cmd = ([hipcc, "-c", str(src), "-o", str(obj)] + cflags + incflags)
        if src.suffix == ".cpp":
            cmd.insert(1, "hip")
            cmd.insert(1, "-x")
        if verbose:
            print("[build]", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)
        objs.append(str(obj))

    ldflags = ([f"-L{p}" for p in lib] +
               ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
                "-ltorch_hip", "-lc10_hip", "-lamdhip64"])
    cmd = [hipcc, "-shared", "-fPIC", *objs, "-o", str(SO_PATH)] + ldflags
    if verbose:
        print("[build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
