"""Build the xaynet_amd._hip extension with hipcc for gfx950 (in-tree).

hipcc compiles both the device code (kernels.hip) and the pybind11 host
module in one shared object. No GPU is needed to build (cross-compile).
"""
import os
import subprocess
import sys
import sysconfig


def build(verbose=True):
    root = os.path.dirname(os.path.abspath(__file__))
    import pybind11

    src = [
        os.path.join(root, "xaynet_amd", "csrc", "gpu", "kernels.hip"),
        os.path.join(root, "xaynet_amd", "csrc", "gpu", "hip_bindings.cpp"),
    ]
    ext = sysconfig.get_config_var("EXT_SUFFIX")
    out = os.path.join(root, "xaynet_amd", f"_hip{ext}")

    # skip if up to date
    if os.path.exists(out) and all(os.path.getmtime(out) > os.path.getmtime(s) for s in src):
        if verbose:
            print(f"_hip up to date: {out}")
        return out

    hipcc = "/opt/rocm/bin/hipcc"
    cmd = [
        hipcc,
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        "-x", "hip", src[0],
        "-x", "hip", src[1],  # host-only file, but hipcc needs the HIP language mode
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        "-o", out,
    ]
    if verbose:
        print(" ".join(cmd))
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build()
