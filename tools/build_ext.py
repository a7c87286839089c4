#!/usr/bin/env python3
"""Build the tfosr_hip_ops extension in-tree with hipcc for gfx950.

Drives hipcc directly (no torch hipify pass — the sources are native HIP):
each .hip kernel TU and the torch binding TU compile with
``hipcc --offload-arch=gfx950`` and link into
``tensorflowonspark_amd/ops/tfosr_hip_ops.so``. The .so travels to GPU boxes
with the repo snapshot; rebuilds are incremental on mtime.
"""

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(REPO, "csrc")
OUT_DIR = os.path.join(REPO, "tensorflowonspark_amd", "ops")
OUT_SO = os.path.join(OUT_DIR, "tfosr_hip_ops.so")
BUILD = os.path.join(REPO, "build", "hip")

SOURCES = ["bn_relu.hip", "softmax_xent.hip", "elementwise.hip",
           "gemm_mfma.hip", "conv3x3_mfma.hip", "conv_wrw_mfma.hip", "conv_wrw2.hip", "maxpool.hip", "tfrecord_codec.cpp", "bindings.cpp"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_flags():
    import torch  # noqa: F401
    import torch.utils.cpp_extension as ce
    incs = ["-I" + p for p in ce.include_paths()]
    incs.append("-I" + sysconfig.get_paths()["include"])
    libs = ["-L" + p for p in ce.library_paths()]
    import torch
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defs = [
        "-D_GLIBCXX_USE_CXX11_ABI={}".format(abi),
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DTORCH_EXTENSION_NAME=tfosr_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ]
    return incs, libs, defs


def build(verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    os.makedirs(OUT_DIR, exist_ok=True)
    incs, libs, defs = torch_flags()
    objs = []
    relink = not os.path.exists(OUT_SO)
    for src in SOURCES:
        path = os.path.join(CSRC, src)
        obj = os.path.join(BUILD, os.path.splitext(src)[0] + ".o")
        objs.append(obj)
        deps = [path, os.path.join(CSRC, "tfosr_common.h")]
        if (os.path.exists(obj)
                and all(os.path.getmtime(obj) > os.path.getmtime(d) for d in deps)):
            continue
        relink = True
        cmd = (["hipcc", "--offload-arch=" + ARCH, "-O3", "-std=c++17",
                "-fPIC", "-c", path, "-o", obj, "-x", "hip"]
               + incs + defs + ["-Wno-deprecated-declarations", "-Wno-macro-redefined"])
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
    if relink:
        cmd = (["hipcc", "-shared", "-fPIC", "-o", OUT_SO] + objs + libs
               + ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
                  "-ltorch_hip", "-lc10_hip", "-lamdhip64"])
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
        print("[build_ext] built", OUT_SO, flush=True)
    else:
        print("[build_ext] up to date:", OUT_SO, flush=True)
    return OUT_SO


INFER_BIN = os.path.join(REPO, "tools", "bin", "tfosr_infer")


def build_infer_cli(verbose=True):
    """Build the standalone C++ inference CLI (libtorch, no Python)."""
    os.makedirs(os.path.dirname(INFER_BIN), exist_ok=True)
    incs, libs, defs = torch_flags()
    src = os.path.join(CSRC, "tfosr_infer.cpp")
    codec_obj = os.path.join(BUILD, "tfrecord_codec.o")
    deps = [src, codec_obj]
    if (os.path.exists(INFER_BIN)
            and all(os.path.getmtime(INFER_BIN) > os.path.getmtime(d)
                    for d in deps)):
        print("[build_ext] up to date:", INFER_BIN, flush=True)
        return INFER_BIN
    import torch.utils.cpp_extension as ce
    rpath = ce.library_paths()[0]
    obj = os.path.join(BUILD, "tfosr_infer.o")
    cc = (["hipcc", "--offload-arch=" + ARCH, "-O3", "-std=c++17", "-fPIC",
           "-c", src, "-o", obj] + incs + defs
          + ["-Wno-deprecated-declarations"])
    # link with g++ (hipcc would treat .o inputs as HIP sources)
    link = (["g++", obj, codec_obj, "-o", INFER_BIN] + libs
            + ["-L/opt/rocm/lib", "-ltorch", "-ltorch_cpu", "-lc10",
               "-ltorch_hip", "-lc10_hip", "-lamdhip64",
               "-Wl,-rpath," + rpath, "-Wl,-rpath,/opt/rocm/lib"])
    for cmd in (cc, link):
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
    print("[build_ext] built", INFER_BIN, flush=True)
    return INFER_BIN


if __name__ == "__main__":
    build()
    build_infer_cli()
    sys.exit(0)
