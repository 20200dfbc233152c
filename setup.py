"""Build script for amgcl_amd native extensions.

Three artifacts, all built in-tree so they travel with the repo snapshot:
  - amgcl_amd/_core.*.so           : CPU setup engine (pybind11 + OpenMP)
  - amgcl_amd/_hip/libamghip.so    : hand-written gfx950 HIP kernels + native
                                     solve driver (hipcc, no torch dependency)
  - amgcl_amd/_capi/libamgclamd_c.so: standalone C API (plain C++/OpenMP)

Usage: python setup.py build_ext --inplace   (or amgcl_amd.build.build_all())
"""
import os
import subprocess
import sys

from setuptools import setup

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from amgcl_amd.build import build_capi_lib, build_core_ext, build_hip_lib  # noqa: E402

if __name__ == "__main__":
    if "build_ext" in sys.argv:
        build_core_ext()
        build_hip_lib()
        build_capi_lib()
    else:
        setup(name="amgcl_amd", version="0.1.0", packages=["amgcl_amd"])
