#!/usr/bin/env python3
"""In-tree build of the horovod_amd native core for MI355X (gfx950).

Compiles every csrc/*.cc and csrc/*.hip with hipcc (host C++ and CDNA4
device code in one toolchain — no hipify, no CUDA shims) and links
horovod_amd/_core.so against libtorch + RCCL.  hipcc cross-compiles gfx950
without a GPU present, so this runs in CPU-only containers too.
"""
import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "csrc")
BUILD = os.path.join(ROOT, "build")
OUT = os.path.join(ROOT, "horovod_amd", "_core.so")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
HIPCC = os.path.join(ROCM, "bin", "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch
    import pybind11
    tdir = os.path.dirname(torch.__file__)
    inc = [
        os.path.join(tdir, "include"),
        os.path.join(tdir, "include", "torch", "csrc", "api", "include"),
        pybind11.get_include(),
        sysconfig.get_paths()["include"],
        os.path.join(ROCM, "include"),
    ]
    lib = os.path.join(tdir, "lib")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return inc, lib, abi


def build(verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    inc, torch_lib, abi = torch_paths()

    sources = sorted(
        os.path.join(CSRC, f)
        for f in os.listdir(CSRC)
        if f.endswith((".cc", ".hip"))
    )
    cxxflags = [
        "-O3", "-std=c++17", "-fPIC",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_core",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-Wno-deprecated-declarations",
    ] + [f"-I{d}" for d in inc]
    hipflags = [f"--offload-arch={ARCH}", "-fno-gpu-rdc"]

    objs = []
    cmds = []
    for src in sources:
        base = os.path.basename(src).rsplit(".", 1)[0]
        obj = os.path.join(BUILD, base + ".o")
        objs.append(obj)
        # skip up-to-date objects (cheap dep check: src + headers mtime)
        hdrs = [os.path.join(CSRC, h) for h in os.listdir(CSRC) if h.endswith(".h")]
        newest_dep = max(os.path.getmtime(p) for p in [src] + hdrs)
        if os.path.exists(obj) and os.path.getmtime(obj) > newest_dep:
            continue
        cmd = [HIPCC, "-c", src, "-o", obj] + cxxflags
        if src.endswith(".hip"):
            cmd += hipflags
        else:
            cmd += ["-x", "c++"]
        cmds.append(cmd)

    def run(cmd):
        if verbose:
            print("[build]", os.path.basename(cmd[2]), flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"compile failed: {' '.join(cmd)}\n{r.stdout}\n{r.stderr}")
        return r

    with ThreadPoolExecutor(max_workers=os.cpu_count() or 4) as ex:
        list(ex.map(run, cmds))

    link = [
        HIPCC, "-shared", "-fPIC", "-o", OUT, *objs,
        f"-L{torch_lib}", f"-Wl,-rpath,{torch_lib}",
        "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
        "-ltorch_hip", "-lc10_hip",
        f"-L{ROCM}/lib", f"-Wl,-rpath,{ROCM}/lib",
        "-lrccl", "-lamdhip64",
    ]
    if verbose:
        print("[build] linking _core.so", flush=True)
    r = subprocess.run(link, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    if verbose:
        print(f"[build] wrote {OUT}", flush=True)
    return OUT


if __name__ == "__main__":
    build()
    sys.exit(0)
