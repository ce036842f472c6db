"""Kernel isolation: direct encode->GPU decode->compare, no engine."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from scanner_amd import _core

def smooth(n, h, w):
    yy, xx = np.mgrid[0:h, 0:w]
    rng = np.random.RandomState(0)
    tex = rng.randint(0, 32, size=(h, w, 3)).astype(np.int32)
    f = np.zeros((n, h, w, 3), np.uint8)
    for i in range(n):
        f[i, :, :, 0] = (xx + 2 * i + tex[:, :, 0]) % 256
        f[i, :, :, 1] = (yy + i + tex[:, :, 1]) % 256
        f[i, :, :, 2] = (xx + yy + 3 * i + tex[:, :, 2]) % 256
    return f

_core.init_memory(1 << 30, 4 << 30, [0])
for (h, w) in [(360, 640), (1080, 1920)]:
    frames = smooth(4, h, w)
    got = _core.svc_gpu_roundtrip(frames, 16, [])
    fg, fr = got.reshape(4, -1), frames.reshape(4, -1)
    print(f"== {h}x{w} direct")
    for i in range(4):
        mism = np.nonzero(fg[i] != fr[i])[0]
        print(f"  frame {i}: " + ("OK" if len(mism) == 0 else
              f"{len(mism)} wrong first@{mism[0]} got={fg[i][mism[0]]} want={fr[i][mism[0]]}"))
    # sparse want exercising null-out interior frames
    got2 = _core.svc_gpu_roundtrip(frames, 16, [3])
    m2 = np.nonzero(got2.reshape(1, -1)[0] != fr[3])[0]
    print("  want=[3]: " + ("OK" if len(m2) == 0 else f"{len(m2)} wrong first@{m2[0]}"))
