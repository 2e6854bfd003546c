"""Build the _hetu_hip extension with hipcc for gfx950 (in-tree .so).

Direct hipcc invocation (no hipify, no CUDA shims): every source is native
HIP/CDNA4. Cross-compiles without a GPU; the built .so travels with the
repo snapshot to the GPU box.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))

SOURCES = [
    "norms.hip",
    "reduce.hip",
    "activations.hip",
    "swiglu.hip",
    "rope.hip",
    "softmax.hip",
    "ce.hip",
    "dropout.hip",
    "embedding.hip",
    "adam.hip",
    "gemm.hip",
    "attention.hip",
    "attention_v2.hip",
    "attention_v3.hip",
    "attention_bwd_v3.hip",
    "attention_bwd_v2.hip",
    "quant.hip",
    "ltgemm.cpp",
    "galvatron_dp.cpp",
    "embed_cache.cpp",
    "dataloader.cpp",
    "bindings.cpp",
]


def _torch_paths():
    import torch
    troot = os.path.dirname(torch.__file__)
    inc = [
        os.path.join(troot, "include"),
        os.path.join(troot, "include", "torch", "csrc", "api", "include"),
    ]
    lib = os.path.join(troot, "lib")
    abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"
    return inc, lib, abi


def build(verbose: bool = True) -> str:
    inc, libdir, abi = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    out = os.path.join(HERE, "_hetu_hip.so")
    srcs = [os.path.join(HERE, s) for s in SOURCES
            if os.path.exists(os.path.join(HERE, s))]
    objs = []
    os.makedirs(os.path.join(HERE, ".build"), exist_ok=True)
    common_flags = [
        "-O3", "-std=c++17", "-fPIC",
        "--offload-arch=gfx950",
        "-DTORCH_EXTENSION_NAME=_hetu_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-fno-gpu-rdc",
        "-Wno-deprecated-declarations",
        "-Wno-unused-result",
        # a non-void function falling off its end produced a silent
        # GPU-side hang (see profiles/r02_capture_replay_bug.md era):
        # make it a hard error
        "-Werror=return-type",
    ] + [f"-I{p}" for p in inc] + [f"-I{py_inc}", f"-I{HERE}"]
    for src in srcs:
        base = os.path.basename(src).rsplit(".", 1)[0]
        obj = os.path.join(HERE, ".build", base + ".o")
        # skip if up to date
        if (os.path.exists(obj)
                and os.path.getmtime(obj) > os.path.getmtime(src)
                and os.path.getmtime(obj) > os.path.getmtime(
                    os.path.join(HERE, "common.h"))):
            objs.append(obj)
            continue
        cmd = ["hipcc", "-c", src, "-o", obj] + common_flags
        if verbose:
            print("[hetu_amd build]", os.path.basename(src))
        subprocess.run(cmd, check=True)
        objs.append(obj)
    link = ["hipcc", "-shared", "-fPIC", "-o", out] + objs + [
        f"-L{libdir}", "-ltorch", "-ltorch_cpu", "-lc10",
        "-ltorch_python", "-lamdhip64", "-lhipblaslt",
        f"-Wl,-rpath,{libdir}",
    ]
    hiplibs = [l for l in ("torch_hip", "c10_hip")
               if os.path.exists(os.path.join(libdir, f"lib{l}.so"))]
    for l in hiplibs:
        link.insert(-1, f"-l{l}")
    if verbose:
        print("[hetu_amd build] linking _hetu_hip.so")
    subprocess.run(link, check=True)
    return out


if __name__ == "__main__":
    build()
