#!/usr/bin/env python3
"""Build the windflow_amd native engine + HIP kernels in-tree.

Outputs:
  windflow_amd/_core.so  — pybind11 engine module (host C++ + HIP kernels)

Host C++ is compiled with g++ (HIP host API via -D__HIP_PLATFORM_AMD__),
device code with hipcc --offload-arch=gfx950.  Objects are cached by mtime
under build/.  Usage: python build_ext.py [--force]
"""
import os
import subprocess
import sys
import sysconfig
import concurrent.futures as cf

ROOT = os.path.dirname(os.path.abspath(__file__))
BUILD = os.path.join(ROOT, "build")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
GFX = os.environ.get("WFA_GFX", "gfx950")

PYINC = sysconfig.get_paths()["include"]


def pybind_includes():
    out = subprocess.check_output([sys.executable, "-m", "pybind11", "--includes"]).decode()
    return [f[2:] for f in out.split() if f.startswith("-I")]


CXXFLAGS = [
    "-O3", "-std=c++20", "-fPIC", "-fvisibility=hidden", "-pthread",
    "-D__HIP_PLATFORM_AMD__", "-DWFA_WITH_HIP",
    f"-I{ROCM}/include", f"-I{PYINC}",
] + [f"-I{p}" for p in pybind_includes()]

HIPFLAGS = [
    "-O3", "-std=c++17", "-fPIC", f"--offload-arch={GFX}",
    "-fvisibility=hidden", "-DWFA_WITH_HIP",
]

ENGINE_SRCS = [
    "csrc/engine/core.cpp",
    "csrc/engine/engine.cpp",
    "csrc/engine/native_logic.cpp",
    "csrc/engine/windows.cpp",
    "csrc/engine/persist.cpp",
    "csrc/engine/gpu_ops.cpp",
    "csrc/engine/gpu_jit.cpp",
    "csrc/engine/bindings.cpp",
]
HIP_SRCS = [
    "csrc/hip/kernels.hip",
    "csrc/hip/sortwin.hip",
]


def newer(a, deps):
    if not os.path.exists(a):
        return True
    at = os.path.getmtime(a)
    return any(os.path.getmtime(d) > at for d in deps if os.path.exists(d))


def headers(dirpath):
    out = []
    for dp, _, fns in os.walk(os.path.join(ROOT, dirpath)):
        out += [os.path.join(dp, f) for f in fns if f.endswith((".hpp", ".h", ".cuh"))]
    return out


def compile_one(src, force):
    full = os.path.join(ROOT, src)
    if not os.path.exists(full):
        return None
    obj = os.path.join(BUILD, src.replace("/", "_") + ".o")
    deps = [full] + headers("csrc")
    if not force and not newer(obj, deps):
        return obj
    if src.endswith(".hip"):
        cmd = [f"{ROCM}/bin/hipcc"] + HIPFLAGS + ["-c", full, "-o", obj]
    else:
        cmd = ["g++"] + CXXFLAGS + ["-c", full, "-o", obj]
    print("  CC", src, flush=True)
    subprocess.check_call(cmd)
    return obj


def build(force=False):
    os.makedirs(BUILD, exist_ok=True)
    srcs = [s for s in ENGINE_SRCS + HIP_SRCS if os.path.exists(os.path.join(ROOT, s))]
    with cf.ThreadPoolExecutor(max_workers=os.cpu_count()) as ex:
        objs = list(ex.map(lambda s: compile_one(s, force), srcs))
    objs = [o for o in objs if o]
    out = os.path.join(ROOT, "windflow_amd", "_core.so")
    if newer(out, objs) or force:
        print("  LINK", out, flush=True)
        # PyTorch-ROCm bundles its own libamdhip64.so (DT_NEEDED without the
        # .so.7 suffix).  Linking ours against /opt/rocm's .so.7 loads TWO
        # HIP runtimes when torch is imported in the same process — which
        # crashes.  Link against torch's copy so exactly one runtime exists.
        torchlib = None
        try:
            import torch
            cand = os.path.join(os.path.dirname(torch.__file__), "lib")
            if os.path.exists(os.path.join(cand, "libamdhip64.so")):
                torchlib = cand
        except Exception:
            pass
        if torchlib:
            # same single-runtime rule applies to RCCL (torch bundles its own)
            hip_link = [f"-L{torchlib}", "-l:libamdhip64.so", "-l:librccl.so", "-l:libhiprtc.so",
                        f"-Wl,-rpath,{torchlib}"]
        else:
            hip_link = [f"-L{ROCM}/lib", "-lamdhip64", "-lrccl", "-lhiprtc",
                        f"-Wl,-rpath,{ROCM}/lib"]
        link = ["g++", "-shared", "-o", out] + objs + hip_link + ["-pthread"]
        subprocess.check_call(link)
    return out


def build_stress(tsan=False):
    """Build the native engine stress harness (csrc/tests/engine_stress.cpp)
    — python-free, HIP-free; with tsan=True it is instrumented for
    ThreadSanitizer (race detection the reference lacks)."""
    os.makedirs(BUILD, exist_ok=True)
    out = os.path.join(BUILD, "engine_stress" + ("_tsan" if tsan else ""))
    srcs = [os.path.join(ROOT, p) for p in
            ["csrc/tests/engine_stress.cpp", "csrc/engine/engine.cpp",
             "csrc/engine/core.cpp", "csrc/engine/native_logic.cpp",
             "csrc/engine/windows.cpp", "csrc/engine/persist.cpp",
             "csrc/engine/gpu_ops.cpp", "csrc/engine/gpu_jit.cpp"]]
    if not newer(out, srcs + headers("csrc")):
        return out
    flags = ["-O1" if tsan else "-O2", "-g", "-std=c++20", "-pthread"]
    if tsan:
        flags += ["-fsanitize=thread"]
    cmd = ["g++"] + flags + srcs + ["-o", out]
    print("  BUILD", out, flush=True)
    subprocess.check_call(cmd)
    return out


if __name__ == "__main__":
    if "--stress" in sys.argv:
        build_stress(tsan="--tsan" in sys.argv)
    else:
        build(force="--force" in sys.argv)
    print("OK")
