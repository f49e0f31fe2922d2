"""In-tree build of the native discovery module (plain C++, no HIP).

The resulting ``_p2p.so`` lives next to this file so it travels with repo
snapshots. Built with g++ + pybind11 headers — no torch dependency.
"""
from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
SRC = PKG_DIR / "csrc" / "p2p.cpp"
SO_PATH = PKG_DIR / "_p2p.so"


def needs_build() -> bool:
    return not SO_PATH.exists() or SRC.stat().st_mtime > SO_PATH.stat().st_mtime


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not needs_build():
        return SO_PATH
    import pybind11

    cmd = [
        "g++", "-O2", "-std=c++17", "-fPIC", "-shared", "-pthread",
        "-I", pybind11.get_include(),
        "-I", sysconfig.get_paths()["include"],
        str(SRC), "-o", str(SO_PATH),
    ]
    if verbose:
        print("[dnet_amd.discovery.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
