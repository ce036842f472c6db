#!/usr/bin/env python3
"""Compile a user C++/HIP op into a plugin .so loadable with
Client.load_op (parity: the reference's user-op .so workflow +
scannerpy/build_flags.py, which exposes the compile/link flags).

Usage: python tools/build_op.py my_op.cpp [-o my_op.so]

The source registers ops/kernels with the SCA_REGISTER_OP /
SCA_REGISTER_KERNEL macros (static registrars run at load time, like the
reference's REGISTER_OP). Link is against the in-tree engine
(scanner_amd/_core.so), so plugins use the same Element/memory/kernel SDK
headers as first-party ops. .hip sources compile for gfx950.
"""
import argparse
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PKG = os.path.join(REPO, "scanner_amd")


def build_flags():
    import sysconfig
    py_inc = sysconfig.get_paths()["include"]
    return {
        "cxx": "/opt/rocm/bin/hipcc",
        "cflags": ["-O3", "-std=c++17", "-fPIC", f"-I{PKG}",
                   f"-I{py_inc}", "-I/opt/rocm/include",
                   "-D__HIP_PLATFORM_AMD__", "-Wno-unused-result"],
        "hipflags": ["--offload-arch=gfx950"],
        "ldflags": ["-shared", "-fPIC", f"-L{PKG}", "-l:_core.so",
                    f"-Wl,-rpath,{PKG}", "-L/opt/rocm/lib", "-lamdhip64"],
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("source")
    ap.add_argument("-o", "--output")
    args = ap.parse_args()
    f = build_flags()
    out = args.output or os.path.splitext(args.source)[0] + ".so"
    is_hip = args.source.endswith((".hip", ".cu"))
    cmd = [f["cxx"]] + (f["hipflags"] if is_hip else ["-x", "c++"]) + \
        f["cflags"] + [args.source] + f["ldflags"] + ["-o", out]
    print(" ".join(cmd), file=sys.stderr)
    subprocess.check_call(cmd)
    print(out)


if __name__ == "__main__":
    main()
