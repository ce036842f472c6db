"""Tutorial 03: temporal ops — stencils (optical flow over [0,1] windows)
and slicing for stateful-op parallelism.
(Parity: examples/tutorials 06_slicing + stencil docs.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp


def main():
    sc = sp.Client(db_path=tempfile.mkdtemp(prefix="sca_tut03_"))
    frames = np.random.RandomState(3).randint(
        0, 255, size=(24, 48, 64, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="raw")

    # dense optical flow (stencil [0, 1]) + per-frame summary stats
    frame = sc.io.Input([video])
    flow = sc.ops.OpticalFlow(frame=frame)
    stats = sc.ops.FlowStats(flow=flow)
    out = sp.NamedStream(sc, "flow_stats")
    sc.run(sc.io.Output(stats, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(list(out.load())) == 24

    # slice a long stream into 4 independent groups for a stateful op
    tab = sc.new_table("seq", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(16)])
    col = sc.io.Input([tab])
    sliced = sc.streams.Slice(col, sc.partitioner.all(4))
    inc = sc.ops.TestIncrement(ignore=sliced)
    unsliced = sc.streams.Unslice(inc)
    out2 = sp.NamedStream(sc, "sliced_out")
    sc.run(sc.io.Output(unsliced, [out2]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(list(out2.load())) == 16
    print("stencil + slicing OK")


if __name__ == "__main__":
    main()
