"""Build the rllm_amd _C HIP extension in-tree for gfx950.

Explicit hipcc invocation (not torch's hipify pipeline): the sources are
native HIP/CDNA4 — nothing to translate. Produces rllm_amd/ops/_C.so next
to this file so the built artifact travels with the repo snapshot to GPU
boxes (a JIT cache under ~/.cache would not).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
BUILD = OPS_DIR / "_build"
SO_PATH = OPS_DIR / "_C.so"

SOURCES = [
    "elementwise.hip",
    "logprob.hip",
    "grpo.hip",
    "adamw.hip",
    "sampling.hip",
    "skinny_gemm.hip",
    "hbl_tuned.hip",
    "attention.hip",
    "bindings.cpp",
]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    from torch.utils import cpp_extension

    includes = cpp_extension.include_paths()
    lib_dir = Path(torch.__file__).parent / "lib"
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return includes, lib_dir, abi


def _needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    for src in SOURCES + ["common.hpp"]:
        p = CSRC / src
        if p.exists() and p.stat().st_mtime > so_mtime:
            return True
    return False


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not _needs_rebuild():
        return SO_PATH

    includes, lib_dir, abi = _torch_paths()
    BUILD.mkdir(exist_ok=True)

    py_include = sysconfig.get_paths()["include"]
    common_flags = [
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    include_flags = [f"-I{p}" for p in includes] + [f"-I{py_include}", f"-I{CSRC}"]

    objs = []
    for src in SOURCES:
        src_path = CSRC / src
        obj_path = BUILD / (src.replace(".", "_") + ".o")
        objs.append(str(obj_path))
        if not force and obj_path.exists() and obj_path.stat().st_mtime > src_path.stat().st_mtime and obj_path.stat().st_mtime > (CSRC / "common.hpp").stat().st_mtime:
            continue
        cmd = ["hipcc", f"--offload-arch={ARCH}", "-c", str(src_path), "-o", str(obj_path)]
        cmd += common_flags + include_flags
        if src.endswith(".cpp"):
            cmd.append("-x")
            cmd.append("c++")
        if verbose:
            print("[rllm_amd.ops.build]", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)

    link_cmd = (
        ["hipcc", "-shared", "-o", str(SO_PATH)]
        + objs
        + [f"-L{lib_dir}", "-ltorch", "-ltorch_python", "-lc10", "-ltorch_hip", "-lc10_hip", "-lamdhip64", "-L/opt/rocm/lib", "-lhipblaslt",
           f"-Wl,-rpath,{lib_dir}"]
    )
    if verbose:
        print("[rllm_amd.ops.build]", " ".join(link_cmd), file=sys.stderr)
    subprocess.run(link_cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {SO_PATH}")
