"""Property-based SVC codec tests (hypothesis): byte-exact roundtrip for
arbitrary shapes and content classes, and GOP-span gather decode through
the engine against a direct frame-index model.

Complements the example-based SVC tests (test_engine_cpu.py) the way the
reference's decoder tests sweep videos of different GOP structure
(tests/videos.cpp): here shape, content statistics (which drive group
bit-widths 0..8), and access pattern are all drawn.
"""
import os
import sys
import tempfile

import numpy as np
from hypothesis import HealthCheck, given, settings, strategies as st

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import scanner_amd as sp  # noqa: E402

_client = None
_serial = [0]


def client():
    global _client
    if _client is None:
        tmp = tempfile.mkdtemp(prefix="scanner_codecprop_")
        _client = sp.Client(db_path=os.path.join(tmp, "db"))
    return _client


def fresh(name):
    """NamedVideoStream keeps an existing table (reference ingest-if-absent
    semantics, storage.py NamedVideoStorage), so every example needs a
    fresh table name."""
    _serial[0] += 1
    return f"{name}_{_serial[0]}"


def gen_frames(content, n, h, w, seed, c=3):
    rng = np.random.RandomState(seed)
    if content == "const":
        return np.full((n, h, w, c), seed % 256, np.uint8)
    if content == "random":  # residual width 8 everywhere
        return rng.randint(0, 256, size=(n, h, w, c)).astype(np.uint8)
    if content == "gradient":  # small deltas -> narrow widths
        base = np.arange(h * w * c, dtype=np.uint32).reshape(h, w, c)
        return np.stack([((base + 3 * f) % 256).astype(np.uint8)
                         for f in range(n)])
    if content == "step":  # mixed: half constant, half noisy
        fr = np.full((n, h, w, c), 50, np.uint8)
        fr[:, h // 2:] = rng.randint(0, 256,
                                     size=(n, h - h // 2, w, c))
        return fr
    raise AssertionError(content)


@settings(max_examples=25, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(n=st.integers(1, 10), h=st.integers(4, 33), w=st.integers(4, 37),
       content=st.sampled_from(["const", "random", "gradient", "step"]),
       seed=st.integers(0, 10), c=st.sampled_from([1, 2, 3, 4]))
def test_svc_roundtrip_property(n, h, w, content, seed, c):
    sc = client()
    frames = gen_frames(content, n, h, w, seed, c)
    video = sp.NamedVideoStream(sc, fresh("cprop"), frames=frames, codec="svc")
    got = np.stack(list(video.load()))
    np.testing.assert_array_equal(got, frames)


@settings(max_examples=12, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(gops=st.integers(1, 4), fpg=st.integers(1, 6),
       fps=st.sampled_from([24.0, 30.0, 60.0]))
def test_mp4_export_ingest_duality(gops, fpg, fps, tmp_path_factory):
    """H.264 ingest -> mp4 export -> probe/re-ingest keeps the full index
    (sample count, keyframe structure, dimensions) for every GOP shape.
    Exercises the writer (mp4_write) against the parser (mp4_parse) as
    inverse functions on the index level."""
    from test_video_ingest import make_annexb

    sc = client()
    tmp = tmp_path_factory.mktemp("mp4prop")
    stream, _, keyframes = make_annexb(gops=gops, frames_per_gop=fpg)
    n = gops * fpg
    p = tmp / "clip.h264"
    p.write_bytes(stream)
    name = fresh("mp4prop")
    r = sc.ingest_video_file(str(p), name)
    assert r["num_frames"] == n
    out_path = str(tmp / "out.mp4")
    sp.NamedVideoStream(sc, name).save_mp4(out_path, fps=fps)
    from scanner_amd import _core
    t = _core.mp4_probe(open(out_path, "rb").read())
    assert len(t["sample_offsets"]) == n
    assert t["keyframe_indices"] == keyframes
    r2 = sc.ingest_video_file(out_path, fresh("mp4prop_re"))
    assert r2["num_frames"] == n
    assert (r2["width"], r2["height"]) == (r["width"], r["height"])


@settings(max_examples=15, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(rows=st.lists(st.integers(0, 19), min_size=1, max_size=12),
       seed=st.integers(0, 5))
def test_svc_gather_property(rows, seed):
    """Gather arbitrary (unsorted, possibly duplicated) rows through the
    engine over an SVC table: decode must produce exactly frames[rows],
    whatever GOP spans the row set touches."""
    sc = client()
    frames = gen_frames("gradient", 20, 16, 16, seed)
    video = sp.NamedVideoStream(sc, fresh("cprop_g"), frames=frames, codec="svc")
    col = sc.io.Input([video])
    g = sc.streams.Gather(col, [rows])
    out = sp.NamedStream(sc, "cprop_g_out")
    sc.run(sc.io.Output(g, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    outs = list(sp.NamedVideoStream(sc, "cprop_g_out").load())
    assert len(outs) == len(rows)
    for k, r in enumerate(rows):
        np.testing.assert_array_equal(outs[k], frames[r], err_msg=str(rows))
