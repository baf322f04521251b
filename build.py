#!/usr/bin/env python3
"""In-tree build for the lux-mi355x native libraries and CLI binaries.

Builds (mtime-incremental):
  lux_amd/liblux_cpu.so   g++    — .lux IO, RMAT, partitioner, CPU engines
  lux_amd/liblux_gpu.so   hipcc  — gfx950 HIP kernels + GPU runtime (C ABI)
  bin/<app>               g++    — Lux-CLI-compatible app drivers (CPU path)

The GPU library is cross-compiled for gfx950 on machines without a GPU
(hipcc needs no device to compile); the .so ships to the GPU box in-tree.
"""
import os
import subprocess
import sys
import glob

ROOT = os.path.dirname(os.path.abspath(__file__))
INC = os.path.join(ROOT, "src", "include")
HIPCC = "/opt/rocm/bin/hipcc"

CPU_SRCS = sorted(glob.glob(os.path.join(ROOT, "src", "core", "*.cpp")))
GPU_SRCS = sorted(glob.glob(os.path.join(ROOT, "src", "gpu", "*.hip")))
CPU_LIB = os.path.join(ROOT, "lux_amd", "liblux_cpu.so")
GPU_LIB = os.path.join(ROOT, "lux_amd", "liblux_gpu.so")

# NOTE: no -fopenmp — loading libgomp before torch breaks ROCm device
# detection in the same process (found via tests/ import bisection).
CXXFLAGS = ["-O2", "-std=c++17", "-fPIC", "-Wall", f"-I{INC}"]
HIPFLAGS = [
    "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC", f"-I{INC}",
    "-Wall", "-Wno-unused-function",
]


def _newer(target, deps):
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    return any(os.path.getmtime(d) > t for d in deps)


def _run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd, cwd=ROOT)


def build_cpu(force=False):
    hdrs = glob.glob(os.path.join(INC, "lux", "*.h"))
    if force or _newer(CPU_LIB, CPU_SRCS + hdrs):
        _run(["g++", "-shared", *CXXFLAGS, *CPU_SRCS, "-o", CPU_LIB])
    return CPU_LIB


def build_gpu(force=False):
    if not GPU_SRCS:
        return None
    hdrs = glob.glob(os.path.join(INC, "lux", "*.h")) + glob.glob(
        os.path.join(ROOT, "src", "gpu", "*.h"))
    if force or _newer(GPU_LIB, GPU_SRCS + hdrs):
        if not os.path.exists(HIPCC):
            raise RuntimeError("hipcc not found at %s" % HIPCC)
        objs = []
        for s in GPU_SRCS:
            obj = s.replace(".hip", ".o")
            if _newer(obj, [s] + hdrs) or force:
                _run([HIPCC, "-c", *HIPFLAGS, s, "-o", obj])
            objs.append(obj)
        _run([HIPCC, "-shared", *HIPFLAGS, *objs, "-o", GPU_LIB])
    return GPU_LIB


def build_apps(force=False):
    """Lux-CLI-compatible standalone binaries: tools/ are CPU (g++ +
    liblux_cpu); apps/ are single-GPU native drivers (hipcc + runtime +
    liblux_gpu). Multi-GPU runs go through the Python RCCL engine."""
    app_srcs = sorted(glob.glob(os.path.join(ROOT, "apps", "*.cpp")))
    tool_srcs = sorted(glob.glob(os.path.join(ROOT, "tools", "*.cpp")))
    runtime = sorted(glob.glob(os.path.join(ROOT, "src", "runtime", "*.cpp")))
    bin_dir = os.path.join(ROOT, "bin")
    os.makedirs(bin_dir, exist_ok=True)
    hdrs = glob.glob(os.path.join(INC, "lux", "*.h")) + glob.glob(
        os.path.join(ROOT, "src", "runtime", "*.h")) + glob.glob(
        os.path.join(ROOT, "apps", "*.h"))
    rpath = f"-Wl,-rpath,{os.path.dirname(CPU_LIB)}"
    out = []
    for s in tool_srcs:
        name = os.path.splitext(os.path.basename(s))[0]
        exe = os.path.join(bin_dir, name)
        if force or _newer(exe, [s, CPU_LIB] + hdrs):
            _run(["g++", *CXXFLAGS, s, CPU_LIB, "-o", exe, rpath])
        out.append(exe)
    if os.path.exists(GPU_LIB):
        for s in app_srcs:
            name = os.path.splitext(os.path.basename(s))[0]
            exe = os.path.join(bin_dir, name)
            if force or _newer(exe, [s, CPU_LIB, GPU_LIB] + runtime + hdrs):
                libdir = os.path.dirname(CPU_LIB)
                _run([HIPCC, *HIPFLAGS, "-Wno-unused-value", s, *runtime,
                      f"-L{libdir}", "-llux_cpu", "-llux_gpu",
                      "-L/opt/rocm/lib", "-lrccl", "-o", exe, rpath])
            out.append(exe)
    return out


def build_all(force=False):
    build_cpu(force)
    build_gpu(force)
    build_apps(force)


if __name__ == "__main__":
    force = "--force" in sys.argv
    if "--cpu" in sys.argv:
        build_cpu(force)
    elif "--gpu" in sys.argv:
        build_gpu(force)
    else:
        build_all(force)
