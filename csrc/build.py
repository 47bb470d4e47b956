#!/usr/bin/env python3
"""In-tree build of the gfx950 HIP extension (_lstm_hip.so).

Direct hipcc invocation (no JIT cache): the built .so lands inside the
package directory so it travels to the GPU box with the repo snapshot.
Cross-compiles fine on a GPU-less machine.
"""
from __future__ import annotations

import hashlib
import json
import os
import subprocess
import sys
import sysconfig

CSRC = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(CSRC)
PKG = os.path.join(REPO, "lstm_tensorspark_amd")
BUILD = os.path.join(CSRC, ".hip_build")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

SOURCES = ["lstm_kernels.hip", "bindings.cpp"]
OUT_SO = os.path.join(PKG, "_lstm_hip.so")


def torch_flags():
    import torch
    from torch.utils import cpp_extension as ce

    includes = [f"-I{p}" for p in ce.include_paths()]
    includes.append(f"-I{sysconfig.get_paths()['include']}")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defines = [
        "-DTORCH_EXTENSION_NAME=_lstm_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DHIPBLAS_V2",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
    ]
    torch_lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    link = [
        f"-L{torch_lib}", f"-Wl,-rpath,{torch_lib}",
        "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
        "-ltorch_hip", "-lc10_hip",
        "-L/opt/rocm/lib", "-Wl,-rpath,/opt/rocm/lib", "-lamdhip64",
    ]
    return includes, defines, link


def digest() -> str:
    h = hashlib.sha256()
    for s in SOURCES + ["build.py"]:
        with open(os.path.join(CSRC, s), "rb") as f:
            h.update(f.read())
    h.update(ARCH.encode())
    # build flavor is part of the identity: otherwise a debug/ASAN .so
    # would pass the up-to-date check for a release build (and a debug
    # session would silently reuse the release .so)
    h.update(os.environ.get("LSTM_TS_BUILD_MODE", "release").encode())
    h.update(os.environ.get("LSTM_TS_ASAN", "0").encode())
    return h.hexdigest()


def run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build(force: bool = False) -> str:
    stamp = os.path.join(BUILD, "stamp.json")
    dig = digest()
    if not force and os.path.exists(OUT_SO) and os.path.exists(stamp):
        try:
            if json.load(open(stamp))["digest"] == dig:
                print(f"extension up to date: {OUT_SO}")
                return OUT_SO
        except Exception:
            pass
    os.makedirs(BUILD, exist_ok=True)
    includes, defines, link = torch_flags()
    common = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
              "-Wno-unused-result"]
    # Debug / sanitizer build (SURVEY.md §5 race-detection row):
    #   LSTM_TS_BUILD_MODE=debug -> -O1 -g + device asserts
    #   LSTM_TS_ASAN=1           -> host AddressSanitizer on the bindings
    if os.environ.get("LSTM_TS_BUILD_MODE") == "debug":
        common = [f"--offload-arch={ARCH}", "-O1", "-g", "-std=c++17",
                  "-fPIC", "-Wno-unused-result", "-DLSTM_TS_DEBUG"]
    if os.environ.get("LSTM_TS_ASAN") == "1":
        common += ["-fsanitize=address", "-shared-libasan"]
        link += ["-fsanitize=address", "-shared-libasan"]
    objs = []
    for src in SOURCES:
        obj = os.path.join(BUILD, os.path.splitext(src)[0] + ".o")
        cmd = [HIPCC, "-c", os.path.join(CSRC, src), "-o", obj] + common
        if src.endswith(".cpp"):
            cmd += includes + defines + ["-x", "hip"]
        run(cmd)
        objs.append(obj)
    run([HIPCC, "-shared", "-fPIC", "-o", OUT_SO] + objs + link)
    json.dump({"digest": dig, "arch": ARCH}, open(stamp, "w"))
    print(f"built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
