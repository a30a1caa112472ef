"""Build libxaynet_ffi.so — the C-ABI participant library (the reference's
xaynet-mobile cdylib equivalent). Links the protocol core + SDK + HTTP client
statically; exports only the xaynet_ffi_* symbols (see
xaynet_amd/csrc/ffi/xaynet_ffi.h)."""
import glob
import os
import subprocess

ROOT = os.path.dirname(os.path.abspath(__file__))
OUT = os.path.join(ROOT, "xaynet_amd", "libxaynet_ffi.so")


def build():
    srcs = sorted(
        s
        for s in glob.glob(os.path.join(ROOT, "xaynet_amd/csrc/*.cpp"))
        + glob.glob(os.path.join(ROOT, "xaynet_amd/csrc/*/*.cpp"))
        if "/gpu/" not in s and "_bindings" not in s
    )
    newest_src = max(os.path.getmtime(s) for s in srcs)
    if os.path.exists(OUT) and os.path.getmtime(OUT) > newest_src:
        return OUT
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", "-o", OUT, *srcs,
           "-pthread", "-lssl", "-lcrypto"]
    subprocess.run(cmd, check=True, cwd=ROOT)
    return OUT


if __name__ == "__main__":
    print(build())
