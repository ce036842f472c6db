"""Tutorial 04: profile a job and export a Chrome trace (open in
chrome://tracing or https://ui.perfetto.dev).
(Parity: examples/tutorials/07_profiling.py.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp


def main():
    db = tempfile.mkdtemp(prefix="sca_tut04_")
    sc = sp.Client(db_path=db)
    frames = np.random.RandomState(4).randint(
        0, 255, size=(32, 64, 80, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    blur = sc.ops.Blur(frame=frame, kernel_size=5)
    hist = sc.ops.Histogram(frame=blur)
    out = sp.NamedStream(sc, "prof_out")
    prof = sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(8, 16),
                  cache_mode=sp.CacheMode.Overwrite)
    stats = prof.statistics()
    for label in sorted(stats, key=lambda k: -stats[k]["total_ms"])[:5]:
        print(f"{label:30s} {stats[label]['total_ms']:8.2f} ms "
              f"x{stats[label]['count']}")
    trace = prof.write_trace(os.path.join(db, "trace.json"))
    print("chrome trace written to", trace)


if __name__ == "__main__":
    main()
