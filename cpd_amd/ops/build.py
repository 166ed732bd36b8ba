"""Ahead-of-time build driver for the cpd_amd native extensions.

Two in-tree shared objects (the reference builds JIT-only via
torch.utils.cpp_extension.load, quant_function.py:10-17; we build AOT so the
.so travels with the repo snapshot and CI can check the build without a GPU):

  * ``_cpd_cpu.so``  — host C++ (g++), always available.
  * ``_cpd_hip.so``  — hand-written HIP/CDNA4 kernels, compiled by hipcc for
    gfx950 only (cross-compiles fine on a machine with no GPU).

Invoked by ``__graft_entry__.build()`` and usable directly:
``python -m cpd_amd.ops.build``.
"""
import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")

GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    incs = list(ce.include_paths()) + [sysconfig.get_paths()["include"]]
    libs = list(ce.library_paths())
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    return incs, libs, abi


def _common_flags(ext_name, incs, abi):
    flags = [
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        f"-DTORCH_EXTENSION_NAME={ext_name}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
    ]
    flags += [f"-I{p}" for p in incs]
    flags += [f"-I{CSRC}"]
    return flags


def _link_flags(libs, hip=False):
    out = []
    for p in libs:
        out += [f"-L{p}", f"-Wl,-rpath,{p}"]
    out += ["-ltorch", "-ltorch_cpu", "-lc10", "-ltorch_python"]
    if hip:
        out += ["-ltorch_hip", "-lc10_hip", "-lamdhip64"]
    return out


def _run(cmd, verbose):
    if verbose:
        print("[cpd-build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def _stale(target, sources):
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    deps = list(sources) + [os.path.join(CSRC, "quant_core.h"), __file__]
    return any(os.path.getmtime(s) > t for s in deps)


def build_cpu(force=False, verbose=True):
    target = os.path.join(HERE, "_cpd_cpu.so")
    srcs = [os.path.join(CSRC, "cpu_ops.cpp")]
    if not force and not _stale(target, srcs):
        return target
    incs, libs, abi = _torch_paths()
    cmd = (
        ["g++"]
        + _common_flags("_cpd_cpu", incs, abi)
        + ["-fopenmp", "-march=native"]
        + srcs
        + _link_flags(libs)
        + ["-o", target]
    )
    _run(cmd, verbose)
    return target


def build_hip(force=False, verbose=True):
    target = os.path.join(HERE, "_cpd_hip.so")
    srcs = sorted(
        os.path.join(CSRC, f)
        for f in os.listdir(CSRC)
        if f.endswith(".hip") or f == "hip_bindings.cpp"
    )
    if not srcs:
        return None
    if not force and not _stale(target, srcs):
        return target
    incs, libs, abi = _torch_paths()
    cmd = (
        ["hipcc", f"--offload-arch={GFX_ARCH}"]
        + _common_flags("_cpd_hip", incs, abi)
        + [
            "-D__HIP_PLATFORM_AMD__=1",
            "-DUSE_ROCM=1",
            "-DHIPBLAS_V2",
            "-fno-gpu-rdc",
            "-Wno-unused-result",
        ]
        + srcs
        + _link_flags(libs, hip=True)
        + ["-o", target]
    )
    _run(cmd, verbose)
    return target


def build_all(force=False, verbose=True):
    cpu = build_cpu(force=force, verbose=verbose)
    hip = build_hip(force=force, verbose=verbose)
    return cpu, hip


if __name__ == "__main__":
    force = "--force" in sys.argv
    cpu, hip = build_all(force=force)
    print("built:", cpu, hip)
