"""In-tree build of the gfx950 extension — explicit hipcc/g++, no torch
cpp_extension machinery and no hipify pass: the .hip files are hand-written
CDNA4 HIP compiled by hipcc as-is; bind.cpp is host-only C++ against
torch's ROCm-native headers.

    python -m pytorch_distributed_train_amd.ops.build
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

import torch

PKG = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG, "csrc")
OBJ = os.path.join(PKG, "_build")
ROCM = os.environ.get("ROCM_HOME", "/opt/rocm")
TORCH_DIR = os.path.dirname(os.path.abspath(torch.__file__))

HIP_SOURCES = ["elementwise.hip", "bn.hip", "pool.hip", "sgd.hip", "ce.hip",
               "gemm.hip", "conv.hip", "ln.hip", "packs.hip", "gemm_strided.hip", "attn.hip"]
CPP_SOURCES = ["bind.cpp"]

SO_NAME = "_hip_ops" + sysconfig.get_config_var("EXT_SUFFIX")


def _includes():
    return [
        f"{TORCH_DIR}/include",
        f"{TORCH_DIR}/include/torch/csrc/api/include",
        f"{ROCM}/include",
        sysconfig.get_paths()["include"],
    ]


def _defines():
    abi = int(torch.compiled_with_cxx11_abi())
    return [
        "__HIP_PLATFORM_AMD__=1", "USE_ROCM=1",
        "TORCH_API_INCLUDE_EXTENSION_H", "TORCH_EXTENSION_NAME=_hip_ops",
        f"_GLIBCXX_USE_CXX11_ABI={abi}",
        # torch headers expect these on ROCm builds
        "CUDA_HAS_FP16=1", "__HIP_NO_HALF_OPERATORS__=1",
        "__HIP_NO_HALF_CONVERSIONS__=1",
    ]


def _needs_build(src: str, obj: str) -> bool:
    if not os.path.exists(obj):
        return True
    omt = os.path.getmtime(obj)
    deps = [src] + [os.path.join(CSRC, h) for h in ("common.h", "mfma_tile.h")]
    return any(os.path.getmtime(d) > omt for d in deps if os.path.exists(d))


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(" ".join(cmd) + "\n" + r.stdout + r.stderr)
        raise RuntimeError(f"build failed: {os.path.basename(cmd[-3])}")


def build(verbose: bool = False) -> None:
    os.makedirs(OBJ, exist_ok=True)
    incs = [f"-I{d}" for d in _includes()]
    defs = [f"-D{d}" for d in _defines()]
    jobs = []
    objs = []
    for f in HIP_SOURCES:
        src = os.path.join(CSRC, f)
        obj = os.path.join(OBJ, f.replace(".hip", ".o"))
        objs.append(obj)
        if _needs_build(src, obj):
            jobs.append([f"{ROCM}/bin/hipcc", "--offload-arch=gfx950", "-O3",
                         "-std=c++17", "-fPIC", *defs, *incs, "-c", src,
                         "-o", obj])
    for f in CPP_SOURCES:
        src = os.path.join(CSRC, f)
        obj = os.path.join(OBJ, f.replace(".cpp", ".o"))
        objs.append(obj)
        if _needs_build(src, obj):
            jobs.append(["g++", "-O3", "-std=c++17", "-fPIC", *defs, *incs,
                         "-c", src, "-o", obj])
    if verbose:
        for j in jobs:
            print("[build]", os.path.basename(j[-3]))
    with ThreadPoolExecutor(max_workers=min(8, max(1, len(jobs)))) as tp:
        list(tp.map(_run, jobs))

    so_path = os.path.join(PKG, SO_NAME)
    if jobs or not os.path.exists(so_path):
        link = ["g++", "-shared", "-o", so_path, *objs,
                f"-L{TORCH_DIR}/lib", f"-L{ROCM}/lib",
                "-Wl,-rpath," + f"{TORCH_DIR}/lib",
                "-Wl,-rpath," + f"{ROCM}/lib",
                "-ltorch", "-ltorch_cpu", "-ltorch_python", "-ltorch_hip",
                "-lc10", "-lc10_hip", "-lamdhip64", "-lrccl"]
        _run(link)
        if verbose:
            print("[build] linked", so_path)

    # import check
    import importlib
    importlib.import_module("pytorch_distributed_train_amd.ops._hip_ops")


if __name__ == "__main__":
    build(verbose=True)
