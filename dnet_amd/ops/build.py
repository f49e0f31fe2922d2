"""In-tree build of the dnet_amd HIP extension for gfx950.

Drives hipcc directly (no hipify, no JIT cache outside the repo): the
resulting ``_C.so`` lives next to this file so it travels with the repo
snapshot to GPU boxes. hipcc cross-compiles gfx950 on machines with no GPU.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
SO_PATH = OPS_DIR / "_C.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _sources() -> list[Path]:
    return sorted(CSRC.glob("*.cpp"))


def _newest_mtime(paths) -> float:
    return max(p.stat().st_mtime for p in paths)


def needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    srcs = _sources() + list(CSRC.glob("*.h"))
    return _newest_mtime(srcs) > SO_PATH.stat().st_mtime


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not needs_build():
        return SO_PATH
    from torch.utils import cpp_extension as ce

    includes = ce.include_paths() + [sysconfig.get_paths()["include"]]
    lib_dirs = ce.library_paths()
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [
        hipcc,
        "-O3",
        "-std=c++17",
        f"--offload-arch={ARCH}",
        "-fPIC",
        "-shared",
        "-fno-gpu-rdc",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        # match the wheel's ABI
        "-D_GLIBCXX_USE_CXX11_ABI=1",
    ]
    for inc in includes:
        cmd += ["-I", inc]
    cmd += [str(s) for s in _sources()]
    for d in lib_dirs:
        cmd += ["-L", d, f"-Wl,-rpath,{d}"]
    cmd += ["-ltorch", "-ltorch_python", "-lc10", "-ltorch_hip", "-lc10_hip",
            "-lamdhip64", "-o", str(SO_PATH)]
    if verbose:
        print("[dnet_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
