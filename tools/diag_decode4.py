import sys, os
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
h, w = 1080, 1920
for n, gop, tag in [(1, 16, "n1"), (2, 16, "n2"), (4, 16, "n4"),
                    (4, 1, "n4-allkey"), (2, 1, "n2-allkey")]:
    frames = smooth(n, h, w)
    got = _core.svc_gpu_roundtrip(frames, gop, [])
    fg, fr = got.reshape(n, -1), frames.reshape(n, -1)
    res = []
    for i in range(n):
        m = np.nonzero(fg[i] != fr[i])[0]
        res.append("OK" if len(m) == 0 else f"f{i}:{len(m)}@{m[0]}")
    print(tag, " ".join(res))
