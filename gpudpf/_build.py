"""In-tree native build driver for gpudpf.

Builds two pybind11 extension modules directly with the system toolchain
(no pip, no JIT cache outside the tree — the built .so files sit inside the
package so they travel to the GPU box with the repo snapshot):

  gpudpf/_core.so  - CPU DPF core (g++)
  gpudpf/_hip.so   - MI355X HIP kernels + runtime (hipcc --offload-arch=gfx950)

Rebuilds are mtime-cached.  `python -m gpudpf._build` forces a build.
"""

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(REPO, "csrc")
PKG = os.path.join(REPO, "gpudpf")

GFX_ARCH = os.environ.get("GPUDPF_GFX_ARCH", "gfx950")


def _pybind_includes():
    import pybind11

    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


def _ext_suffix():
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _needs_build(target, sources):
    if not os.path.exists(target):
        return True
    tmt = os.path.getmtime(target)
    deps = list(sources)
    for d in (os.path.join(CSRC, "core"), os.path.join(CSRC, "hip")):
        for f in os.listdir(d):
            if f.endswith((".h", ".hip", ".cc")):
                deps.append(os.path.join(d, f))
    return any(os.path.getmtime(s) > tmt for s in deps if os.path.exists(s))


def _run(cmd):
    print("[gpudpf build]", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build_core(force=False):
    sources = [
        os.path.join(CSRC, "core", "dpf_core.cc"),
        os.path.join(CSRC, "core", "prf.cc"),
        os.path.join(CSRC, "core", "prf_avx2.cc"),
        os.path.join(CSRC, "core", "aes128.cc"),
        os.path.join(CSRC, "core", "core_bindings.cc"),
    ]
    target = os.path.join(PKG, "_core" + _ext_suffix())
    if not force and not _needs_build(target, sources):
        return target
    cmd = (
        ["g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-march=native",
         "-fvisibility=hidden", "-pthread"]
        + ["-I" + i for i in _pybind_includes()]
        + sources
        + ["-o", target]
    )
    _run(cmd)
    return target


def build_hip(force=False):
    sources = [
        os.path.join(CSRC, "hip", "dpf_kernels.hip"),
        os.path.join(CSRC, "hip", "gemm128.hip"),
        os.path.join(CSRC, "hip", "gemm_u32.hip"),
        os.path.join(CSRC, "hip", "gemm_stream.hip"),
        os.path.join(CSRC, "hip", "hip_bindings.cc"),
        os.path.join(CSRC, "core", "aes128.cc"),
    ]
    target = os.path.join(PKG, "_hip" + _ext_suffix())
    if not force and not _needs_build(target, sources):
        return target
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    extra = os.environ.get("GPUDPF_HIP_EXTRA_FLAGS", "").split()
    cmd = (
        [hipcc, "-O3", "-std=c++17", "-shared", "-fPIC",
         "--offload-arch=" + GFX_ARCH, "-fvisibility=hidden",
         "-I" + os.path.join(CSRC, "core"),
         "-x", "hip"]
        + extra
        + ["-I" + i for i in _pybind_includes()]
        + sources
        + ["-o", target]
    )
    _run(cmd)
    return target


def build(force=False):
    build_core(force=force)
    build_hip(force=force)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
