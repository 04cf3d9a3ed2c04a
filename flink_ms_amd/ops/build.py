"""Direct hipcc build of the flink_ms_amd HIP extension (no hipify).

The kernels are native HIP/CDNA4 source; torch's CUDAExtension pipeline runs
its CUDA->HIP rewriter over them, which we neither need nor want.  This
module drives hipcc explicitly:

  1. hipcc --offload-arch=gfx950 -c  <kernels>.hip        (no torch headers)
  2. hipcc -c bindings.cpp  with torch/pybind include paths (host-only TU)
  3. hipcc -shared -fPIC  -> flink_ms_amd/_hip_ops.so  (in-tree, so the
     built artifact travels with the source snapshot to GPU boxes)

Used by ``__graft_entry__.build()`` and ``python -m flink_ms_amd.ops.build``.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent.parent
CSRC = PKG_DIR / "ops" / "csrc"
SO_PATH = PKG_DIR / "_hip_ops.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = ["als_kernels.hip", "svm_kernels.hip", "serve_kernels.hip",
                  "debug_kernels.hip"]
KVSERVER_SRC = "kvserver.cpp"  # serving/csrc: host-only C++ (Netty parity)


def _hipcc() -> str:
    return os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _torch_paths():
    import torch

    troot = Path(torch.__file__).resolve().parent
    includes = [
        troot / "include",
        troot / "include" / "torch" / "csrc" / "api" / "include",
    ]
    return troot, includes


def _run(cmd, verbose):
    if verbose:
        print("+", " ".join(map(str, cmd)), flush=True)
    subprocess.run([str(c) for c in cmd], check=True)


def build(verbose: bool = True, force: bool = False) -> Path:
    """Compile and link the extension in-tree; returns the .so path."""
    troot, tincs = _torch_paths()
    objdir = PKG_DIR / "ops" / "_build"
    objdir.mkdir(exist_ok=True)

    kv_src = PKG_DIR / "serving" / "csrc" / KVSERVER_SRC
    srcs = [CSRC / s for s in KERNEL_SOURCES] + [CSRC / "bindings.cpp",
                                                 kv_src]
    deps = srcs + [CSRC / "common.hip.h", CSRC / "als_kernels_device.inc",
            Path(__file__)]
    if not force and SO_PATH.exists():
        so_mtime = SO_PATH.stat().st_mtime
        if all(d.stat().st_mtime < so_mtime for d in deps):
            return SO_PATH

    objs = []
    common = ["-O3", "-std=c++17", "-fPIC"]
    for src in KERNEL_SOURCES:
        obj = objdir / (Path(src).stem + ".o")
        _run([_hipcc(), f"--offload-arch={ARCH}", *common, "-c",
              CSRC / src, "-o", obj], verbose)
        objs.append(obj)

    # bindings: host-only TU (torch + pybind11 APIs; no device code)
    py_inc = sysconfig.get_paths()["include"]
    abi = "1"
    try:
        import torch
        abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"
    except Exception:
        pass
    bobj = objdir / "bindings.o"
    _run([
        _hipcc(), *common,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        *[f"-I{p}" for p in tincs],
        f"-I{py_inc}",
        "-x", "c++", "-c", CSRC / "bindings.cpp", "-o", bobj,
    ], verbose)
    objs.append(bobj)

    # native KvState query server (host-only C++, pybind11 only)
    kobj = objdir / "kvserver.o"
    _run([
        _hipcc(), *common,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        *[f"-I{p}" for p in tincs],
        f"-I{py_inc}",
        "-x", "c++", "-c", kv_src, "-o", kobj,
    ], verbose)
    objs.append(kobj)

    _run([
        _hipcc(), "-shared", "-fPIC", *objs, "-o", SO_PATH,
        f"-L{troot / 'lib'}", "-ltorch", "-ltorch_cpu", "-ltorch_python",
        "-lc10", f"-Wl,-rpath,{troot / 'lib'}",
    ], verbose)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {SO_PATH}")
