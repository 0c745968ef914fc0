"""Build the _pio_hip extension in-tree with hipcc for gfx950.

No hipify, no CUDA path: the .hip sources are native CDNA4 code; hipcc
compiles host + device and links against libtorch's ROCm build. The .so is
written next to this file so it travels with the repo snapshot to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
SO_PATH = os.path.join(OPS_DIR, "_pio_hip.so")
SOURCES = ["bindings.cpp", "als_kernels.hip", "topk_kernels.hip",
           "topk_mfma.hip"]
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    root = os.path.dirname(torch.__file__)
    return {
        "includes": [
            os.path.join(root, "include"),
            os.path.join(root, "include", "torch", "csrc", "api", "include"),
        ],
        "libdir": os.path.join(root, "lib"),
        "abi": int(torch._C._GLIBCXX_USE_CXX11_ABI),
    }


def needs_build() -> bool:
    if not os.path.exists(SO_PATH):
        return True
    so_mtime = os.path.getmtime(SO_PATH)
    for s in SOURCES:
        if os.path.getmtime(os.path.join(CSRC, s)) > so_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> str:
    if not force and not needs_build():
        return SO_PATH
    tp = _torch_paths()
    py_include = sysconfig.get_paths()["include"]
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [
        hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-shared",
        "-DTORCH_EXTENSION_NAME=_pio_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={tp['abi']}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-Wno-deprecated-declarations", "-Wno-unused-result",
        f"-I{py_include}",
    ]
    for inc in tp["includes"]:
        cmd.append(f"-I{inc}")
    cmd += [os.path.join(CSRC, s) for s in SOURCES]
    cmd += [
        f"-L{tp['libdir']}", f"-Wl,-rpath,{tp['libdir']}",
        "-ltorch", "-ltorch_cpu", "-ltorch_python", "-ltorch_hip",
        "-lc10", "-lc10_hip", "-lamdhip64",
        "-o", SO_PATH,
    ]
    if verbose:
        print("[ops.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
