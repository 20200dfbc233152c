"""In-tree builders for the native extensions.

- _core: pybind11 CPU extension (OpenMP), compiled with the system g++.
- libamghip.so: pure HIP library compiled with hipcc for gfx950 only
  (no torch headers; the Python side talks to it through ctypes with raw
  device pointers from torch tensors).

Both .so files land inside the package so the gpurun snapshot carries them.
"""
import os
import subprocess
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
REPO_DIR = os.path.dirname(PKG_DIR)
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
GPU_ARCH = os.environ.get("AMGCL_AMD_ARCH", "gfx950")


def _newer(target, sources):
    if not os.path.exists(target):
        return False
    t = os.path.getmtime(target)
    return all(os.path.getmtime(s) <= t for s in sources)


def build_core_ext(verbose=True):
    import pybind11

    src = os.path.join(PKG_DIR, "csrc", "core", "core.cpp")
    ext_suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out = os.path.join(PKG_DIR, "_core" + ext_suffix)
    if _newer(out, [src, __file__]):
        return out
    inc_py = sysconfig.get_paths()["include"]
    inc_pb = pybind11.get_include()
    cmd = [
        "g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-fopenmp",
        "-march=x86-64-v3", "-fvisibility=hidden",
        f"-I{inc_py}", f"-I{inc_pb}",
        src, "-o", out,
    ]
    if verbose:
        print("[amgcl_amd.build] " + " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return out


def build_hip_lib(verbose=True):
    hip_dir = os.path.join(PKG_DIR, "csrc", "hip")
    srcs = sorted(
        os.path.join(hip_dir, f) for f in os.listdir(hip_dir) if f.endswith(".hip")
    )
    if not srcs:
        return None
    out_dir = os.path.join(PKG_DIR, "_hip")
    os.makedirs(out_dir, exist_ok=True)
    out = os.path.join(out_dir, "libamghip.so")
    if _newer(out, srcs + [__file__]):
        return out
    cmd = [
        HIPCC, f"--offload-arch={GPU_ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", "-ffast-math", "-fopenmp",
        # A/B experiment knobs, e.g. "-DAMGCL_NO_SWIZZLE -DAMGCL_NO_NT"
        *os.environ.get("AMGCL_HIP_DEFINES", "").split(),
        *srcs, "-o", out,
    ]
    if verbose:
        print("[amgcl_amd.build] " + " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return out


def build_capi_lib(verbose=True):
    """C API shared library (plain C++/OpenMP, no Python dependency) —
    parity with the reference's C-callable lib/ + Fortran entry points."""
    src = os.path.join(PKG_DIR, "csrc", "capi", "amgcl_amd_c.cpp")
    out_dir = os.path.join(PKG_DIR, "_capi")
    os.makedirs(out_dir, exist_ok=True)
    out = os.path.join(out_dir, "libamgclamd_c.so")
    if _newer(out, [src, __file__]):
        return out
    cmd = [
        "g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-fopenmp",
        "-march=x86-64-v3", src, "-o", out,
    ]
    if verbose:
        print("[amgcl_amd.build] " + " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return out


def build_all(verbose=True):
    build_core_ext(verbose)
    build_hip_lib(verbose)
    build_capi_lib(verbose)


if __name__ == "__main__":
    build_all()
