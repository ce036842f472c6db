"""App: shot-boundary detection (parity: the reference's flagship demo —
examples/apps/shot_detection — histogram-difference cuts).

Pipeline: decode -> per-frame RGB histogram (HIP kernel on GPU, CPU here)
-> a stateful Python op that flags frames whose histogram L1-distance to
the previous frame exceeds an adaptive threshold. Demonstrates: first-party
GPU-capable ops feeding a @register_python_op class kernel with state,
stencils not required (state carries the previous histogram), and result
post-processing on the client.
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

import scanner_amd as sp
from scanner_amd import FrameType  # noqa: F401  (annotation type)

tmp = tempfile.mkdtemp(prefix="scanner_tut10_")
sc = sp.Client(db_path=os.path.join(tmp, "db"))

# Synthesize a clip with 3 hard cuts: 4 "scenes" of very different content.
rng = np.random.RandomState(0)
h, w, scene_len = 120, 160, 12
scenes = []
for base in (20, 200, 90, 150):
    scene = np.full((scene_len, h, w, 3), base, np.uint8)
    scene += rng.randint(0, 12, size=scene.shape).astype(np.uint8)
    scenes.append(scene)
clip = np.concatenate(scenes)
expected_cuts = [scene_len * k for k in (1, 2, 3)]

video = sp.NamedVideoStream(sc, "shots_clip", frames=clip, codec="svc")


@sp.register_python_op(name="ShotScore", bounded_state=True, warmup=1)
class ShotScore(sp.Kernel):
    """L1 distance between consecutive normalized histograms; the previous
    histogram is op state (bounded, warmup 1: task boundaries recompute
    one row of history)."""

    def __init__(self, config, **kw):
        super().__init__(config)
        self.prev = None

    def reset(self):
        self.prev = None

    def execute(self, hist: bytes) -> bytes:
        cur = np.frombuffer(hist, np.uint32).astype(np.float64)
        cur = cur / max(1.0, cur.sum())
        score = 0.0 if self.prev is None else float(
            np.abs(cur - self.prev).sum())
        self.prev = cur
        return np.float64(score).tobytes()


frame = sc.io.Input([video])
hist = sc.ops.Histogram(frame=frame)
score = sc.ops.ShotScore(hist=hist)
out = sp.NamedStream(sc, "shot_scores")
sc.run(sc.io.Output(score, [out]), sp.PerfParams.manual(8, 16),
       cache_mode=sp.CacheMode.Overwrite)

scores = np.array([np.frombuffer(b, np.float64)[0] for b in out.load()])
# adaptive threshold: mean + 3 sigma of the score stream
thresh = scores.mean() + 3 * scores.std()
cuts = [int(i) for i in np.nonzero(scores > thresh)[0]]
print(f"scores: n={len(scores)} mean={scores.mean():.4f} "
      f"thresh={thresh:.4f}")
print("detected cuts at frames:", cuts)
assert cuts == expected_cuts, (cuts, expected_cuts)
print("shot detection OK")
