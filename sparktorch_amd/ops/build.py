"""In-tree build of the _sparkhip extension for gfx950.

Drives hipcc directly for the device code (pure HIP — no hipify, no CUDA
shims) and g++ for the torch-binding TU, linking one in-tree shared object
``sparktorch_amd/ops/_sparkhip<ext>.so`` so the built artifact travels with
the repo snapshot to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

import torch
from torch.utils import cpp_extension

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
BUILD_DIR = os.path.join(OPS_DIR, "build")

HIP_SOURCES = ["elementwise.hip", "loss.hip", "gemm.hip", "conv.hip", "norm.hip", "conv_nhwc.hip", "conv_implicit.hip"]
CPP_SOURCES = ["bindings.cpp"]

ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _ext_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(OPS_DIR, "_sparkhip" + suffix)


def _needs_build(out: str, sources) -> bool:
    if not os.path.exists(out):
        return True
    out_m = os.path.getmtime(out)
    for s in sources:
        if os.path.getmtime(s) > out_m:
            return True
    return False


def _run(cmd, verbose):
    if verbose:
        print(" ".join(cmd), flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            "build command failed:\n%s\nstdout:\n%s\nstderr:\n%s"
            % (" ".join(cmd), r.stdout[-4000:], r.stderr[-8000:])
        )
    return r


def build_extension(
    verbose: bool = False,
    force: bool = False,
    out: str = None,
    build_dir: str = None,
) -> str:
    """Build the extension.  Default: in-tree, mtime-gated.  ``out`` /
    ``build_dir`` redirect the artifact and objects elsewhere (the from-clean
    build check uses a temp dir so it always compiles every TU without
    touching the committed in-tree binary)."""
    in_tree = out is None
    out = out or _ext_path()
    build_dir = build_dir or BUILD_DIR
    srcs = [os.path.join(CSRC, s) for s in HIP_SOURCES + CPP_SOURCES] + [
        os.path.join(CSRC, "common.h")
    ]
    if in_tree and not force and not _needs_build(out, srcs):
        return out

    os.makedirs(build_dir, exist_ok=True)
    torch_inc = cpp_extension.include_paths()
    torch_lib = cpp_extension.library_paths()[0]
    py_inc = sysconfig.get_paths()["include"]
    abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"

    objs = []
    hipcc = os.path.join(ROCM, "bin", "hipcc")
    for s in HIP_SOURCES:
        obj = os.path.join(build_dir, s.replace(".hip", ".o"))
        _run(
            [
                hipcc,
                "--offload-arch=" + ARCH,
                "-O3",
                "-std=c++17",
                "-fPIC",
                "-ffast-math",
                "-c",
                os.path.join(CSRC, s),
                "-o",
                obj,
            ],
            verbose,
        )
        objs.append(obj)

    for s in CPP_SOURCES:
        obj = os.path.join(build_dir, s.replace(".cpp", ".o"))
        cmd = (
            ["g++", "-O2", "-std=c++17", "-fPIC", "-c", os.path.join(CSRC, s), "-o", obj]
            + ["-I" + p for p in torch_inc + [py_inc, os.path.join(ROCM, "include")]]
            + [
                "-D_GLIBCXX_USE_CXX11_ABI=" + abi,
                "-D__HIP_PLATFORM_AMD__=1",
                "-DUSE_ROCM=1",
                "-DHIPBLAS_V2",
                "-DTORCH_EXTENSION_NAME=_sparkhip",
                "-DTORCH_API_INCLUDE_EXTENSION_H",
            ]
        )
        _run(cmd, verbose)
        objs.append(obj)

    link = (
        ["g++", "-shared", "-o", out]
        + objs
        + [
            "-L" + torch_lib,
            "-lc10",
            "-lc10_hip",
            "-ltorch",
            "-ltorch_cpu",
            "-ltorch_hip",
            "-ltorch_python",
            "-L" + os.path.join(ROCM, "lib"),
            "-lamdhip64",
            "-Wl,-rpath," + torch_lib,
            "-Wl,-rpath," + os.path.join(ROCM, "lib"),
        ]
    )
    _run(link, verbose)
    return out


if __name__ == "__main__":
    path = build_extension(verbose=True, force="--force" in sys.argv)
    print("built", path)
