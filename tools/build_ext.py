"""Build the draco_amd._hip_ops extension with hipcc for gfx950, in-tree.

Direct hipcc invocation (no hipify, no nvcc shims): the source is HIP-native.
The resulting .so lands at draco_amd/_hip_ops.<abi>.so so it travels with the repo
snapshot to GPU boxes (it is .gitignored — history stays source-only).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(REPO, "draco_amd", "ops", "csrc", "draco_kernels.hip")


def so_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(REPO, "draco_amd", f"_hip_ops{suffix}")


def needs_build() -> bool:
    out = so_path()
    if not os.path.exists(out):
        return True
    return os.path.getmtime(SRC) > os.path.getmtime(out)


def build(verbose: bool = True, force: bool = False) -> str:
    out = so_path()
    if not force and not needs_build():
        return out
    tdir = os.path.dirname(torch.__file__)
    t_inc = os.path.join(tdir, "include")
    py_inc = sysconfig.get_paths()["include"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    cmd = [
        "hipcc",
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        SRC,
        "-o", out,
        f"-I{t_inc}",
        f"-I{os.path.join(t_inc, 'torch', 'csrc', 'api', 'include')}",
        f"-I{py_inc}",
        "-D__HIP_PLATFORM_AMD__",
        "-DUSE_ROCM",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        f"-L{os.path.join(tdir, 'lib')}",
        "-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python", "-lamdhip64",
        f"-Wl,-rpath,{os.path.join(tdir, 'lib')}",
        "-Wno-unused-result",
    ]
    if verbose:
        print("[build_ext]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(so_path())
