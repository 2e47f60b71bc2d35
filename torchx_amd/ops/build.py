"""In-tree build of the torchx_amd HIP extension (gfx950 only).

Drives hipcc directly on native HIP sources — no hipify, no CUDA shims.
The resulting `_hip_ops.so` lives inside the package so it travels with the
repo snapshot to GPU boxes (a JIT cache under ~/.cache would not).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
SO_PATH = PKG_DIR / "_hip_ops.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

SOURCES = [
    "elementwise.hip",
    "rmsnorm.hip",
    "adamw.hip",
    "cross_entropy.hip",
    "attention.hip",
    "decode_attn.hip",
    "gemv.hip",
    "fp8_cast.hip",
    "probe.hip",
    "bindings.cpp",
]


def _torch_paths():
    import torch

    troot = Path(torch.__file__).parent
    return troot / "include", troot / "lib"


def _needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    for src in CSRC.iterdir():
        if src.suffix in (".hip", ".cpp", ".h") and src.stat().st_mtime > so_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not _needs_build():
        return SO_PATH
    tinc, tlib = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    objs = []
    build_dir = PKG_DIR / "build"
    build_dir.mkdir(exist_ok=True)

    common_flags = [
        "-O3",
        "-fPIC",
        "-std=c++17",
        f"--offload-arch={ARCH}",
        "-DNDEBUG",
        "-D__HIP_PLATFORM_AMD__",
        "-DUSE_ROCM",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-fno-gpu-rdc",
        "-Wno-deprecated-declarations",
        f"-I{tinc}",
        f"-I{tinc}/torch/csrc/api/include",
        f"-I{py_inc}",
    ]

    procs = []
    for src in SOURCES:
        obj = build_dir / (src.replace(".", "_") + ".o")
        objs.append(obj)
        src_path = CSRC / src
        if obj.exists() and obj.stat().st_mtime > src_path.stat().st_mtime \
                and obj.stat().st_mtime > (CSRC / "common.h").stat().st_mtime \
                and not force:
            continue
        cmd = ["hipcc", "-c", str(src_path), "-o", str(obj)] + common_flags
        if src.endswith(".cpp"):
            cmd.append("-x")
            cmd.append("hip")  # bindings still need hip headers/stream types
        if verbose:
            print(f"[torchx_amd.ops.build] hipcc -c {src}", file=sys.stderr)
        procs.append((src, subprocess.Popen(cmd, stderr=subprocess.PIPE)))

    failed = False
    for src, p in procs:
        _, err = p.communicate()
        if p.returncode != 0:
            failed = True
            print(f"--- build failed: {src} ---\n{err.decode()}", file=sys.stderr)
    if failed:
        raise RuntimeError("hipcc compilation failed")

    link = (
        ["hipcc", "-shared", "-fPIC", "-o", str(SO_PATH)]
        + [str(o) for o in objs]
        + [f"-L{tlib}", "-ltorch", "-ltorch_python", "-ltorch_hip", "-lc10",
           "-lc10_hip", f"-Wl,-rpath,{tlib}"]
    )
    if verbose:
        print("[torchx_amd.ops.build] linking _hip_ops.so", file=sys.stderr)
    subprocess.run(link, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
