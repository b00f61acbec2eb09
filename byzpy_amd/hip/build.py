"""In-tree build of the gfx950 HIP extension with explicit hipcc.

No torch hipify pass, no CUDA shims: the sources are native HIP/CDNA4 and
hipcc (clang++) compiles them directly for --offload-arch=gfx950. The
resulting byzpy_amd/_hip_ops.so travels with the repo snapshot to GPU
boxes (JIT caches under ~/.cache would not).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent.parent  # byzpy_amd/
CSRC = PKG_DIR / "hip" / "csrc"
OUT_SO = PKG_DIR / "_hip_ops.so"
BUILD_DIR = PKG_DIR.parent / "build" / "hip"

SOURCES = ["colsel.hip", "rsel.hip", "rowops.hip", "gram.hip", "krumsel.hip",
           "subsets.hip", "attacks.hip", "bind.cpp"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch

    root = Path(torch.__file__).parent
    inc = root / "include"
    return [
        str(inc),
        str(inc / "torch" / "csrc" / "api" / "include"),
    ], str(root / "lib")


def _hipcc() -> str:
    return os.environ.get("HIPCC", "hipcc")


def needs_rebuild() -> bool:
    if not OUT_SO.exists():
        return True
    so_mtime = OUT_SO.stat().st_mtime
    for src in SOURCES + ["common.h", "build.py"]:
        p = CSRC / src if (CSRC / src).exists() else Path(__file__)
        if p.exists() and p.stat().st_mtime > so_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_rebuild():
        if verbose:
            print(f"[byzpy_amd] {OUT_SO} is up to date")
        return OUT_SO
    includes, torch_lib = _torch_paths()
    includes.append(sysconfig.get_paths()["include"])
    BUILD_DIR.mkdir(parents=True, exist_ok=True)

    common = [
        _hipcc(),
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DUSE_ROCM=1",
        "-DGLIBCXX_USE_CXX11_ABI=1",
        "-Wno-deprecated-declarations",
    ] + [f"-I{p}" for p in includes]

    objs = []

    def compile_one(src: str) -> str:
        obj = str(BUILD_DIR / (Path(src).stem + ".o"))
        cmd = common + ["-c", str(CSRC / src), "-o", obj]
        if src.endswith(".hip"):
            cmd.insert(1, "-xhip")
        if verbose:
            print("[byzpy_amd] compiling", src)
        subprocess.run(cmd, check=True)
        return obj

    with ThreadPoolExecutor(max_workers=len(SOURCES)) as ex:
        objs = list(ex.map(compile_one, SOURCES))

    link = (
        common
        + ["-shared", "-o", str(OUT_SO)]
        + objs
        + [
            f"-L{torch_lib}",
            f"-Wl,-rpath,{torch_lib}",
            "-ltorch",
            "-ltorch_cpu",
            "-ltorch_hip",
            "-lc10",
            "-lc10_hip",
            "-ltorch_python",
            "-lamdhip64",
        ]
    )
    if verbose:
        print("[byzpy_amd] linking", OUT_SO.name)
    subprocess.run(link, check=True)
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
