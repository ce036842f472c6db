#!/usr/bin/env python3
"""Build scanner_amd/_core.so in-tree with hipcc (gfx950) via ninja.

hipcc compiles both host .cpp files and .hip device files; the built .so is
committed-adjacent (git-ignored) and travels to GPU boxes with the repo
snapshot. No JIT cache, no site-packages install.
"""
import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(REPO, "scanner_amd")
BUILD = os.path.join(REPO, "build")

CPP_SOURCES = [
    "csrc/memory.cpp",
    "csrc/storage.cpp",
    "csrc/metadata.cpp",
    "csrc/dag/sampler.cpp",
    "csrc/dag/analysis.cpp",
    "csrc/ops/registry.cpp",
    "csrc/ops/stdlib_cpu.cpp",
    "csrc/ops/python_kernel.cpp",
    "csrc/ops/resnet50.cpp",
    "csrc/ops/pose.cpp",
    "csrc/ops/image_encoder.cpp",
    "csrc/ops/detector.cpp",
    "csrc/ops/files_source.cpp",
    "csrc/engine/table_io.cpp",
    "csrc/engine/executor.cpp",
    "csrc/video/svc_cpu.cpp",
    "csrc/video/span_cache.cpp",
    "csrc/video/h264.cpp",
    "csrc/video/mp4.cpp",
    "csrc/video/ingest.cpp",
    "csrc/core_module.cpp",
]
HIP_SOURCES = [
    "kernels/image_ops.hip",
    "kernels/svc_codec.hip",
    "kernels/color.hip",
    "kernels/optflow.hip",
    "kernels/gemm_mfma.hip",
    "kernels/dnn_ops.hip",
]


def main():
    os.makedirs(BUILD, exist_ok=True)
    py_inc = sysconfig.get_paths()["include"]
    pybind_inc = subprocess.check_output(
        [sys.executable, "-m", "pybind11", "--includes"]).decode().strip()
    hipcc = "/opt/rocm/bin/hipcc"
    arch = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
    common = (f"-O3 -std=c++17 -fPIC -I{py_inc} {pybind_inc} "
              f"-I{SRC} -I/opt/rocm/include -D__HIP_PLATFORM_AMD__ "
              f"-Wno-unused-result")
    hipflags = f"--offload-arch={arch}"

    hip_srcs = [s for s in HIP_SOURCES
                if os.path.exists(os.path.join(SRC, s))]

    lines = [
        f"hipcc = {hipcc}",
        f"cflags = {common}",
        f"hipflags = {hipflags}",
        "rule cxx",
        "  command = $hipcc -x c++ $cflags -MD -MF $out.d -c $in -o $out",
        "  depfile = $out.d",
        "  deps = gcc",
        "rule hip",
        "  command = $hipcc $hipflags $cflags -MD -MF $out.d -c $in -o $out",
        "  depfile = $out.d",
        "  deps = gcc",
        "rule link",
        "  command = $hipcc -shared -fPIC $in -o $out -L/opt/rocm/lib "
        "-lamdhip64 -lz",
    ]
    objs = []
    for s in CPP_SOURCES:
        obj = os.path.join(BUILD, s.replace("/", "_") + ".o")
        lines.append(f"build {obj}: cxx {os.path.join(SRC, s)}")
        objs.append(obj)
    for s in hip_srcs:
        obj = os.path.join(BUILD, s.replace("/", "_") + ".o")
        lines.append(f"build {obj}: hip {os.path.join(SRC, s)}")
        objs.append(obj)
    out_so = os.path.join(SRC, "_core.so")
    lines.append(f"build {out_so}: link {' '.join(objs)}")
    lines.append(f"default {out_so}")
    with open(os.path.join(BUILD, "build.ninja"), "w") as f:
        f.write("\n".join(lines) + "\n")
    subprocess.check_call(["ninja", "-C", BUILD])
    print(f"built {out_so}")


if __name__ == "__main__":
    main()
