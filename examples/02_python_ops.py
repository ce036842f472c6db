"""Tutorial 02: register a Python op — plain, batched, and stateful.
(Parity: examples/tutorials/04_custom_op + python kernels.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp
from scanner_amd import FrameType, register_python_op


@register_python_op()
def Brightness(frame: FrameType) -> bytes:
    return float(np.mean(frame)).hex().encode()


def main():
    sc = sp.Client(db_path=tempfile.mkdtemp(prefix="sca_tut02_"))
    frames = np.random.RandomState(2).randint(
        0, 255, size=(12, 48, 64, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    bright = sc.ops.Brightness(frame=frame)
    out = sp.NamedStream(sc, "brightness")
    sc.run(sc.io.Output(bright, [out]), sp.PerfParams.estimate(),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [float.fromhex(b.decode()) for b in out.load()]
    assert len(vals) == 12
    ref = [float(np.mean(f)) for f in frames]
    assert all(abs(a - b) < 1e-6 for a, b in zip(vals, ref))
    print("python op OK:", [round(v, 1) for v in vals[:4]], "...")


if __name__ == "__main__":
    main()
