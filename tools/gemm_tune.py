#!/usr/bin/env python3
"""GEMM micro-benchmark at the DNN ops' real shapes (run on GPU via
gpurun). Prints ms and TFLOP/s per shape."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from scanner_amd import _core

if not _core.have_gpu():
    print("gemm_tune needs a GPU (run via gpurun)", file=sys.stderr)
    sys.exit(2)

# (label, M, N, K): ResNet-50 @ batch 16 and Pose @ batch 8 hot shapes
SHAPES = [
    ("rn.conv1   ", 16 * 112 * 112, 64, 192),
    ("rn.l1.c2   ", 16 * 56 * 56, 64, 576),
    ("rn.l1.c3   ", 16 * 56 * 56, 256, 64),
    ("rn.l2.c2   ", 16 * 28 * 28, 128, 1152),
    ("rn.l3.c2   ", 16 * 14 * 14, 256, 2304),
    ("rn.l4.c2   ", 16 * 7 * 7, 512, 4608),   # split-K path
    ("rn.l4.c2b32", 32 * 7 * 7, 512, 4608),
    ("rn.l4.ds   ", 32 * 7 * 7, 2048, 1024),
    ("rn.fc      ", 32, 1024, 2048),
    ("pose.b2    ", 8 * 184 * 184, 64, 576),
    ("pose.b4    ", 8 * 92 * 92, 128, 1152),
    ("pose.stage ", 8 * 46 * 46, 128, 1728),
    ("smallk.c3  ", 16 * 56 * 56, 256, 64),   # small-K M-walk path
    ("smallk.wide", 16 * 56 * 56, 128, 64),
    ("square4k   ", 4096, 4096, 4096),
    ("square8k   ", 8192, 8192, 8192),
]

for label, M, N, K in SHAPES:
    ms = _core.gemm_bench(M, N, K, 20, True)
    tf = 2.0 * M * N * K / (ms * 1e-3) / 1e12
    print(f"{label} M={M:<8} N={N:<5} K={K:<5} {ms:8.3f} ms  {tf:7.1f} TF/s",
          flush=True)
