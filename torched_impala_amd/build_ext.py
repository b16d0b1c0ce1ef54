"""In-tree extension builds (no JIT cache — the built .so files live next
to the package so they travel to the GPU box with the repo snapshot).

Two artifacts:
  torched_impala_amd/_impala_ring.so  — host C++ (g++): lock-free MPMC queue
  torched_impala_amd/_impala_hip.so   — gfx950 HIP (hipcc): kernels + pinning

Driven by __graft_entry__.build() and `python setup.py build_ext`.
hipcc cross-compiles gfx950 without a GPU present.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")


def _torch_paths():
    import torch
    import torch.utils.cpp_extension as C
    includes = list(C.include_paths())
    lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    return includes, lib, abi


def _common_flags(name: str):
    includes, torch_lib, abi = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    flags = ["-O3", "-std=c++17", "-fPIC", "-shared",
             f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
             f"-DTORCH_EXTENSION_NAME={name}",
             "-DTORCH_API_INCLUDE_EXTENSION_H"]
    flags += [f"-I{p}" for p in includes + [py_inc]]
    link = [f"-L{torch_lib}", "-ltorch", "-ltorch_cpu", "-lc10",
            "-ltorch_python", f"-Wl,-rpath,{torch_lib}"]
    return flags, link, torch_lib


def _needs_build(out: str, sources) -> bool:
    if not os.path.exists(out):
        return True
    t_out = os.path.getmtime(out)
    return any(os.path.getmtime(s) > t_out for s in sources
               if os.path.exists(s))


def _run(cmd):
    print("[build_ext]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build_ring(force: bool = False) -> str:
    out = os.path.join(PKG_DIR, "_impala_ring.so")
    sources = [os.path.join(CSRC, "ring.cpp"),
               os.path.join(CSRC, "ring_core.h")]
    if force or _needs_build(out, sources):
        flags, link, _ = _common_flags("_impala_ring")
        _run(["g++"] + flags + [sources[0]] + link + ["-o", out])
    return out


def build_actor(force: bool = False) -> str:
    """Native CPU actor inference: -mavx2 -mfma (EPYC-safe baseline)."""
    out = os.path.join(PKG_DIR, "_impala_actor.so")
    sources = [os.path.join(CSRC, "actor_net.cpp")]
    if force or _needs_build(out, sources):
        flags, link, _ = _common_flags("_impala_actor")
        _run(["g++", "-mavx2", "-mfma", "-funroll-loops"] + flags + sources
             + link + ["-o", out])
    return out


HIP_SOURCES = ["hip_ext.cpp", "vtrace_kernels.hip", "lstm_kernels.hip",
               "optim_kernels.hip", "conv_kernels.hip", "linear_kernels.hip",
               "deep_kernels.hip", "shallow_mfma.hip", "shallow_mfma_bwd.hip",
               "pack_kernels.hip"]


def build_probe(force: bool = False) -> str:
    """Standalone torch-free GPU validator/bench (deep_probe.cpp) — runs in
    ~1 s of box time, used for kernel bring-up without the torch import
    cost."""
    out = os.path.join(CSRC, "deep_probe")
    sources = [os.path.join(CSRC, "deep_probe.cpp"),
               os.path.join(CSRC, "deep_kernels.hip"),
               os.path.join(CSRC, "shallow_mfma.hip"),
               os.path.join(CSRC, "shallow_mfma_bwd.hip")]
    if force or _needs_build(out, sources):
        _run(["hipcc", "--offload-arch=gfx950", "-O2", "-Wno-unused-value",
              "-x", "hip"] + sources + ["-o", out])
    return out


def build_hip(force: bool = False) -> str:
    out = os.path.join(PKG_DIR, "_impala_hip.so")
    sources = [os.path.join(CSRC, s) for s in HIP_SOURCES
               if os.path.exists(os.path.join(CSRC, s))]
    if not sources:
        raise FileNotFoundError("no HIP sources in csrc/")
    if force or _needs_build(out, sources):
        flags, link, _ = _common_flags("_impala_hip")
        rocm_lib = "/opt/rocm/lib"
        link += [f"-L{rocm_lib}", "-lamdhip64", "-lc10_hip", "-ltorch_hip",
                 f"-Wl,-rpath,{rocm_lib}"]
        cmd = (["hipcc", "--offload-arch=gfx950", "-DUSE_ROCM",
                "-D__HIP_PLATFORM_AMD__", "-fno-gpu-rdc",
                "-Wno-deprecated-declarations", "-Wno-unused-result",
                "-x", "hip"]  # treat .cpp/.hip alike as HIP
               + flags + sources + link + ["-o", out])
        _run(cmd)
    return out


def build_all(force: bool = False):
    build_ring(force)
    build_actor(force)
    try:
        build_hip(force)
    except FileNotFoundError:
        print("[build_ext] HIP sources not present yet; skipped")
    build_probe(force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
