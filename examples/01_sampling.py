"""Tutorial 01: sparse sampling — stride, ranges, gather. Only the GOP
spans covering requested frames are read and decoded.
(Parity: examples/tutorials 01/02 sampling + stride.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp


def main():
    sc = sp.Client(db_path=tempfile.mkdtemp(prefix="sca_tut01_"))
    frames = np.random.RandomState(1).randint(
        0, 255, size=(60, 64, 80, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="svc")

    frame = sc.io.Input([video])
    every_tenth = sc.streams.Stride(frame, [10])
    hist = sc.ops.Histogram(frame=every_tenth)
    out = sp.NamedStream(sc, "hist_strided")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.estimate(),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(list(out.load())) == 6

    frame = sc.io.Input([video])
    some = sc.streams.Gather(frame, [[0, 7, 33, 59]])
    hist = sc.ops.Histogram(frame=some)
    out = sp.NamedStream(sc, "hist_gather")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.estimate(),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(list(out.load())) == 4
    print("sampling OK")


if __name__ == "__main__":
    main()
