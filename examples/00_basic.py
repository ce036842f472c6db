"""Tutorial 00: ingest a clip, compute a histogram per frame, read results.
(Parity: examples/tutorials/00_basic.py in the reference.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp


def main():
    db = tempfile.mkdtemp(prefix="sca_tut00_")
    sc = sp.Client(db_path=db)

    # A synthetic clip stands in for a video file (no network here); with
    # real footage you would pass frames decoded by any source, or ingest
    # an SVC/raw stream directly.
    frames = np.random.RandomState(0).randint(
        0, 255, size=(30, 120, 160, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "tutorial_clip", frames=frames,
                                codec="svc")

    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "tutorial_hist")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.estimate(),
           cache_mode=sp.CacheMode.Overwrite)

    rows = list(out.load())
    assert len(rows) == 30
    h0 = np.frombuffer(rows[0], np.uint32).reshape(3, 256)
    print("first-frame histogram sums:", h0.sum(axis=1))


if __name__ == "__main__":
    main()
