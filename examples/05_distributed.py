"""Tutorial 05: distributed execution — gRPC master + two worker
processes pulling tasks, with fault tolerance.
(Parity: examples/tutorials/08_distributed.py.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp
from scanner_amd.master import MasterServer
from scanner_amd.worker import start_worker


def main():
    db = tempfile.mkdtemp(prefix="sca_tut05_")
    master = MasterServer(db)
    workers = [start_worker(master.addr, db) for _ in range(2)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        frames = np.random.RandomState(5).randint(
            0, 255, size=(24, 48, 64, 3), dtype=np.uint8)
        video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="raw")
        frame = sc.io.Input([video])
        hist = sc.ops.Histogram(frame=frame)
        out = sp.NamedStream(sc, "dist_hist")
        sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)
        assert len(list(out.load())) == 24
        print("distributed job OK across", len(workers), "workers")
    finally:
        for w in workers:
            w.shutdown()
        master.shutdown()


if __name__ == "__main__":
    main()
