"""Source/Sink/Enumerator SDK (csrc/ops/source.h — parity:
scanner/api/source.h:68, sink.h, enumerator.h:49-55): feed the engine from
a directory of files, write results to files, bypassing tables in both
directions."""
import os

import numpy as np
import pytest

import scanner_amd as sp
from scanner_amd import _core


def _write_blobs(d, n, prefix="f"):
    paths = []
    for i in range(n):
        p = os.path.join(d, f"{prefix}{i:03d}.dat")
        with open(p, "wb") as f:
            f.write(bytes([i % 256]) * (10 + 3 * i))
        paths.append(p)
    return paths


def test_registries_present():
    assert "Files" in _core.registered_sources()
    assert "Files" in _core.registered_sinks()


def test_files_source_to_table(sc, tmp_path):
    paths = _write_blobs(str(tmp_path), 9)
    col = sc.sources.Files(paths)
    cat = sc.ops.ConcatBytes(inputs=[col, col])
    out = sp.NamedStream(sc, "fs_out")
    sc.run(sc.io.Output(cat, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 9
    for i, r in enumerate(rows):
        blob = bytes([i % 256]) * (10 + 3 * i)
        assert r == blob + blob


def test_files_source_with_sampler(sc, tmp_path):
    paths = _write_blobs(str(tmp_path), 20)
    col = sc.sources.Files(paths)
    sampled = sc.streams.Stride(col, [3])
    cat = sc.ops.ConcatBytes(inputs=[sampled])
    out = sp.NamedStream(sc, "fss_out")
    sc.run(sc.io.Output(cat, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 7  # ceil(20/3)
    for k, r in enumerate(rows):
        i = 3 * k
        assert r == bytes([i % 256]) * (10 + 3 * i)


def test_files_sink(sc, tmp_path):
    frames = np.random.RandomState(0).randint(
        0, 255, size=(10, 32, 32, 3)).astype(np.uint8)
    video = sp.NamedVideoStream(sc, "fsink_in", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir)
    sink = sc.sinks.Files(hist, out_dir, ext="hist")
    sc.run(sink, sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    for i in range(10):
        p = os.path.join(out_dir, f"c0_{i}.hist")
        assert os.path.exists(p), p
        got = np.fromfile(p, dtype=np.uint32).reshape(3, 256)
        expect = np.stack([np.bincount(frames[i][:, :, c].ravel(),
                                       minlength=256)
                           for c in range(3)]).astype(np.uint32)
        np.testing.assert_array_equal(got, expect)


def test_files_source_to_files_sink_multijob(sc, tmp_path):
    """Two jobs end-to-end with per-job source paths and sink dirs and no
    table anywhere."""
    d0, d1 = str(tmp_path / "a"), str(tmp_path / "b")
    o0, o1 = str(tmp_path / "oa"), str(tmp_path / "ob")
    for d in (d0, d1, o0, o1):
        os.makedirs(d)
    p0 = _write_blobs(d0, 4, "x")
    p1 = _write_blobs(d1, 6, "y")
    col = sc.sources.Files([p0, p1])
    cat = sc.ops.ConcatBytes(inputs=[col])
    sink = sc.sinks.Files(cat, [o0, o1])
    sc.run(sink, sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(os.listdir(o0)) == 4
    assert len(os.listdir(o1)) == 6
    got = open(os.path.join(o1, "c0_5.bin"), "rb").read()
    assert got == bytes([5]) * (10 + 15)


def test_unknown_source_rejected(sc):
    with pytest.raises(Exception, match="unknown source"):
        sc.sources.custom("NoSuchSource", [{}])


def test_files_source_missing_file(sc, tmp_path):
    col = sc.sources.Files([str(tmp_path / "nope.dat")])
    cat = sc.ops.ConcatBytes(inputs=[col])
    out = sp.NamedStream(sc, "fsm_out")
    with pytest.raises(Exception, match="cannot open"):
        sc.run(sc.io.Output(cat, [out]), sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
