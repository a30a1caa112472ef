"""In-tree build of the xaynet_amd native extensions.

Two extensions:
  - xaynet_amd._core : pure C++ protocol core (crypto, masking, messages,
    coordinator + SDK state machines). pybind11, no torch dependency.
  - xaynet_amd._hip  : HIP/CDNA4 kernels + GPU aggregation engine for
    MI355X (gfx950), built with torch.utils.cpp_extension (hipcc), linked
    against torch for tensor interop. Built only when requested (it is
    compiled by __graft_entry__.build() / `python setup.py build_ext`).

Build in-tree: `python setup.py build_ext --inplace` — the resulting .so files
travel to the GPU box with the repo snapshot.
"""
import glob
import os
import sys

from setuptools import setup

import pybind11
from pybind11.setup_helpers import Pybind11Extension, build_ext

ROOT = os.path.dirname(os.path.abspath(__file__))

core_sources = sorted(
    src
    for src in glob.glob("xaynet_amd/csrc/*.cpp") + glob.glob("xaynet_amd/csrc/*/*.cpp")
    # gpu/ builds separately with hipcc (build_hip.py); ffi/ builds as a
    # standalone C-ABI shared library (build_ffi.py)
    if "/gpu/" not in src and "/ffi/" not in src
)

ext_modules = [
    Pybind11Extension(
        "xaynet_amd._core",
        core_sources,
        cxx_std=17,
        extra_compile_args=["-O3", "-fvisibility=hidden", "-g0"],
        libraries=["ssl", "crypto"],  # REST TLS transport (rest/tls.cpp)
    ),
]

setup(
    name="xaynet_amd",
    version="0.1.0",
    description="MI355X-native masked federated learning framework (PET protocol)",
    packages=["xaynet_amd", "xaynet_amd.server", "xaynet_amd.ops", "xaynet_amd.parallel", "xaynet_sdk"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": build_ext},
)
