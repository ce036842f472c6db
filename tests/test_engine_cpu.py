"""End-to-end CPU engine tests (parity role: tests/py_test.py in the
reference — real in-process engine, temp storage, no mocks)."""
import numpy as np
import pytest

import scanner_amd as sp
from conftest import make_smooth_video, make_video


def ref_histogram(frame):
    h = np.zeros((3, 256), dtype=np.uint32)
    for c in range(3):
        h[c] = np.bincount(frame[:, :, c].ravel(), minlength=256)
    return h


def test_histogram_end_to_end(sc):
    frames = make_video(n=30)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "clip_hist")
    sink = sc.io.Output(hist, [out])
    sc.run(sink, sp.PerfParams.manual(8, 16),
           cache_mode=sp.CacheMode.Overwrite)

    rows = list(out.load())
    assert len(rows) == 30
    for i, blob in enumerate(rows):
        got = np.frombuffer(blob, dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(frames[i]))


def test_video_roundtrip_raw(sc):
    frames = make_video(n=10)
    video = sp.NamedVideoStream(sc, "rt", frames=frames, codec="raw")
    got = np.stack(list(video.load()))
    np.testing.assert_array_equal(got, frames)


def test_resize(sc):
    frames = make_video(n=6, h=32, w=40)
    video = sp.NamedVideoStream(sc, "rs", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    small = sc.ops.Resize(frame=frame, width=20, height=16)
    out = sp.NamedStream(sc, "rs_out")
    sc.run(sc.io.Output(small, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    outs = list(sp.NamedVideoStream(sc, "rs_out").load())
    assert len(outs) == 6
    assert outs[0].shape == (16, 20, 3)


def test_multi_job(sc):
    vids = [make_video(n=8, seed=s) for s in range(3)]
    streams = [sp.NamedVideoStream(sc, f"mj{i}", frames=v, codec="raw")
               for i, v in enumerate(vids)]
    frame = sc.io.Input(streams)
    hist = sc.ops.Histogram(frame=frame)
    outs = [sp.NamedStream(sc, f"mj{i}_hist") for i in range(3)]
    sc.run(sc.io.Output(hist, outs), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    for i, o in enumerate(outs):
        rows = list(o.load())
        assert len(rows) == 8
        got = np.frombuffer(rows[3], dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(vids[i][3]))


def test_multi_column_output(sc):
    frames = make_video(n=6)
    video = sp.NamedVideoStream(sc, "mc", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    small = sc.ops.Resize(frame=frame, width=16, height=16)
    out = sp.NamedStream(sc, "mc_out")
    sc.run(sc.io.Output([hist, small], [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    info = sc.table_info("mc_out")
    names = [n for n, _ in info["columns"]]
    assert names == ["histogram", "frame"]
    assert info["num_rows"] == 6


def test_stride_sampling(sc):
    frames = make_video(n=20)
    video = sp.NamedVideoStream(sc, "st", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    sampled = sc.streams.Stride(frame, [3])
    hist = sc.ops.Histogram(frame=sampled)
    out = sp.NamedStream(sc, "st_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 7  # ceil(20/3)
    for k, blob in enumerate(rows):
        got = np.frombuffer(blob, dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(frames[k * 3]))


def test_gather_range_samplers(sc):
    frames = make_video(n=25)
    video = sp.NamedVideoStream(sc, "sm", frames=frames, codec="raw")
    # Gather
    frame = sc.io.Input([video])
    rows_wanted = [0, 3, 3, 17, 24]
    g = sc.streams.Gather(frame, [rows_wanted])
    out = sp.NamedStream(sc, "sm_g")
    sc.run(sc.io.Output(sc.ops.Histogram(frame=g), [out]),
           sp.PerfParams.manual(4, 8), cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == len(rows_wanted)
    for k, blob in enumerate(rows):
        got = np.frombuffer(blob, dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got,
                                      ref_histogram(frames[rows_wanted[k]]))
    # Range
    frame = sc.io.Input([video])
    r = sc.streams.Range(frame, [(5, 15)])
    out2 = sp.NamedStream(sc, "sm_r")
    sc.run(sc.io.Output(sc.ops.Histogram(frame=r), [out2]),
           sp.PerfParams.manual(4, 8), cache_mode=sp.CacheMode.Overwrite)
    assert len(list(out2.load())) == 10


def test_space_repeat_null(sc):
    n = 6
    tab = sc.new_table("ints", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    spaced = sc.streams.RepeatNull(col, [3])
    out = sp.NamedStream(sc, "spaced")
    sc.run(sc.io.Output(spaced, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == n * 3
    for i, rbytes in enumerate(rows):
        if i % 3 == 0:
            assert int.from_bytes(rbytes, "little") == i // 3
        else:
            assert rbytes is None


def test_space_repeat(sc):
    n = 4
    tab = sc.new_table("ints2", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    spaced = sc.streams.Repeat(col, [2])
    out = sp.NamedStream(sc, "rep")
    sc.run(sc.io.Output(spaced, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    assert vals == [0, 0, 1, 1, 2, 2, 3, 3]


def test_bounded_state(sc):
    n = 20
    tab = sc.new_table("bs", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    inc = sc.ops.TestIncrement(ignore=col)
    out = sp.NamedStream(sc, "bs_out")
    # tiny io packets force warmup recomputation across tasks
    sc.run(sc.io.Output(inc, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    # warmup=0 bounded state: state resets at each task start, so each
    # io-packet of 4 rows yields input + (1,2,3,4)
    expect = [i + (i % 4) + 1 for i in range(n)]
    assert vals == expect


def test_unbounded_state(sc):
    n = 12
    tab = sc.new_table("us", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    inc = sc.ops.TestIncrementUnbounded(ignore=col)
    out = sp.NamedStream(sc, "us_out")
    sc.run(sc.io.Output(inc, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    # unbounded state recomputes from row 0 in every task: value = i + i+1
    assert vals == [2 * i + 1 for i in range(n)]


def test_stencil_wider_than_packet(sc):
    frames = make_video(n=12)
    video = sp.NamedVideoStream(sc, "stn", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    flow = sc.ops.OpticalFlow(frame=frame)
    out = sp.NamedStream(sc, "stn_out")
    # io_packet=2 < stencil reach across packets
    sc.run(sc.io.Output(flow, [out]), sp.PerfParams.manual(1, 2),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(sp.NamedVideoStream(sc, "stn_out").load())
    assert len(rows) == 12
    assert rows[0].shape == (48, 64, 2)  # dense per-pixel flow


def test_slice_unslice(sc):
    n = 16
    tab = sc.new_table("sl", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    sliced = sc.streams.Slice(col, sc.partitioner.all(4))
    inc = sc.ops.TestIncrement(ignore=sliced)
    unsliced = sc.streams.Unslice(inc)
    out = sp.NamedStream(sc, "sl_out")
    sc.run(sc.io.Output(unsliced, [out]), sp.PerfParams.manual(8, 16),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    # state resets every 4-row slice group: value = i + (i%4)+1
    assert vals == [i + (i % 4) + 1 for i in range(n)]


def test_overlapping_slices_single_task(sc):
    n = 10
    tab = sc.new_table("ov", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    # overlapping windows [0,6) and [4,10)
    sliced = sc.streams.Slice(
        col, sc.partitioner.ranges([(0, 6), (4, 10)]))
    inc = sc.ops.TestIncrement(ignore=sliced)
    unsliced = sc.streams.Unslice(inc)
    out = sp.NamedStream(sc, "ov_out")
    sc.run(sc.io.Output(unsliced, [out]), sp.PerfParams.manual(8, 16),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    # group 0: rows 0..5 -> i + (i+1); group 1: rows 4..9 -> (4+i) + (i+1)
    expect = [i + i + 1 for i in range(6)] + [4 + i + i + 1 for i in range(6)]
    assert vals == expect


def test_overlapping_slice_with_pergroup_sampling(sc):
    """Mirror of reference test_overlapping_slice (py_test.py:361-375)."""
    frames = make_video(n=35)
    video = sp.NamedVideoStream(sc, "ovs", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    sliced = sc.streams.Slice(
        frame, sc.partitioner.strided_ranges([(0, 15), (5, 25), (15, 35)], 1))
    sampled = sc.streams.Range(frame=None, ranges=None) if False else \
        sc.streams.Range(sliced, [sp.SliceList([
            {"start": 0, "end": 10},
            {"start": 5, "end": 15},
            {"start": 5, "end": 15},
        ])])
    unsliced = sc.streams.Unslice(sampled)
    out = sp.NamedStream(sc, "ovs_out")
    sc.run(sc.io.Output(unsliced, [out]), sp.PerfParams.manual(8, 16),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(sp.NamedVideoStream(sc, "ovs_out").load())
    assert len(rows) == 30
    # group 1 covers source rows 5..25, its range picks local 5..15
    # => source rows 10..20; group 2 covers 15..35 picking local 5..15
    # => source rows 20..30
    np.testing.assert_array_equal(rows[0], frames[0])
    np.testing.assert_array_equal(rows[10], frames[10])
    np.testing.assert_array_equal(rows[20], frames[20])
    np.testing.assert_array_equal(rows[29], frames[29])
    np.testing.assert_array_equal(rows[19], frames[19])


def test_cache_mode(sc):
    frames = make_video(n=4)
    video = sp.NamedVideoStream(sc, "cm", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "cm_out")
    sink = sc.io.Output(hist, [out])
    sc.run(sink, sp.PerfParams.manual(4, 8))
    with pytest.raises(sp.ScannerException):
        frame = sc.io.Input([video])
        sink2 = sc.io.Output(sc.ops.Histogram(frame=frame), [out])
        sc.run(sink2, sp.PerfParams.manual(4, 8))
    # Ignore: no-op
    frame = sc.io.Input([video])
    sink3 = sc.io.Output(sc.ops.Histogram(frame=frame), [out])
    assert sc.run(sink3, sp.PerfParams.manual(4, 8),
                  cache_mode=sp.CacheMode.Ignore) is None


def test_profile(sc):
    frames = make_video(n=8)
    video = sp.NamedVideoStream(sc, "pf", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "pf_out")
    prof = sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
                  cache_mode=sp.CacheMode.Overwrite)
    stats = prof.statistics()
    assert any(k.startswith("op:Histogram") for k in stats)
    trace = prof.write_trace(str(sc._db_path) + "/trace.json")
    import json
    with open(trace) as f:
        data = json.load(f)
    assert len(data["traceEvents"]) > 0


def test_svc_codec_roundtrip(sc):
    frames = make_video(n=40, h=40, w=56)
    video = sp.NamedVideoStream(sc, "svc_rt", frames=frames, codec="svc")
    got = np.stack(list(video.load()))
    np.testing.assert_array_equal(got, frames)
    # sparse seek patterns exercise keyframe-aligned GOP spans
    for rows in ([0], [17], [39], [3, 18, 35], [16], [15, 16, 17]):
        got = list(video.load(rows=rows))
        for k, r in enumerate(rows):
            np.testing.assert_array_equal(got[k], frames[r])


def test_svc_compression_ratio(sc):
    # smooth temporal content must compress well
    from conftest import make_smooth_video
    frames = make_smooth_video(n=64, h=64, w=64)
    sp.NamedVideoStream(sc, "svc_cr", frames=frames, codec="svc")
    info = sc.table_info("svc_cr")
    import os
    db = sc._db_path
    tdir = os.path.join(db, "tables", str(info["id"]))
    total = sum(os.path.getsize(os.path.join(tdir, f))
                for f in os.listdir(tdir))
    raw = frames.nbytes
    assert total < raw * 0.7, f"svc stream {total} vs raw {raw}"


def test_histogram_over_svc(sc):
    frames = make_video(n=24)
    video = sp.NamedVideoStream(sc, "svc_h", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "svc_h_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 24
    for i, blob in enumerate(rows):
        got = np.frombuffer(blob, dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(frames[i]))


def test_svc_strided_decode(sc):
    frames = make_video(n=30)
    video = sp.NamedVideoStream(sc, "svc_s", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    sampled = sc.streams.Stride(frame, [7])
    hist = sc.ops.Histogram(frame=sampled)
    out = sp.NamedStream(sc, "svc_s_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 5
    for k, blob in enumerate(rows):
        got = np.frombuffer(blob, dtype=np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(frames[k * 7]))


def make_textured_pair(h=96, w=128, dx=2, dy=1, seed=3):
    """Frame pair with known integer translation: band-limited random
    texture (LK needs gradients at every scale)."""
    from scipy.ndimage import gaussian_filter
    rng = np.random.RandomState(seed)
    base = gaussian_filter(rng.rand(h, w) * 255, sigma=2.0)
    base = ((base - base.min()) / (np.ptp(base) + 1e-9) * 255).astype(np.uint8)
    f0 = np.stack([base] * 3, axis=-1)
    f1 = np.roll(f0, (dy, dx), axis=(0, 1))
    return np.stack([f0, f1])


def test_optical_flow_translation(sc):
    dx, dy = 2, 1
    frames = make_textured_pair(dx=dx, dy=dy)
    video = sp.NamedVideoStream(sc, "of", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    flow = sc.ops.OpticalFlow(frame=frame)
    out = sp.NamedStream(sc, "of_out")
    sc.run(sc.io.Output(flow, [out]), sp.PerfParams.manual(2, 2),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(sp.NamedVideoStream(sc, "of_out").load())
    assert len(rows) == 2
    f = rows[0]
    assert f.shape == (96, 128, 2) and f.dtype == np.float32
    # interior flow recovers the translation (frame1 = frame0 shifted by
    # (dy,dx), so I1(x+dx,y+dy)=I0(x,y) => LK converges to u=(dx,dy))
    interior = f[16:-16, 16:-16]
    err = np.abs(interior - np.array([dx, dy], np.float32))
    assert np.median(err[..., 0]) < 0.25, np.median(err[..., 0])
    assert np.median(err[..., 1]) < 0.25, np.median(err[..., 1])


def test_flow_stats(sc):
    frames = make_textured_pair(dx=2, dy=1)
    video = sp.NamedVideoStream(sc, "fs", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    flow = sc.ops.OpticalFlow(frame=frame)
    stats = sc.ops.FlowStats(flow=flow)
    out = sp.NamedStream(sc, "fs_out")
    sc.run(sc.io.Output(stats, [out]), sp.PerfParams.manual(2, 2),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 2
    s = np.frombuffer(rows[0], np.float32)
    assert s.shape == (4,)
    # mean |u| ~ 2, mean |v| ~ 1 for the (2,1) translation pair
    assert 1.0 < s[0] < 3.0 and 0.4 < s[1] < 2.0
    assert s[2] >= s[0] and s[3] >= s[1]  # max >= mean


def test_crop(sc):
    frames = make_video(n=4, h=40, w=60)
    video = sp.NamedVideoStream(sc, "cr", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    crop = sc.ops.Crop(frame=frame, x=10, y=5, width=32, height=24)
    out = sp.NamedStream(sc, "cr_out")
    sc.run(sc.io.Output(crop, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    outs = list(sp.NamedVideoStream(sc, "cr_out").load())
    assert len(outs) == 4
    for i, o in enumerate(outs):
        np.testing.assert_array_equal(o, frames[i][5:29, 10:42])


def test_color_convert(sc):
    frames = make_video(n=3, h=24, w=32)
    video = sp.NamedVideoStream(sc, "ccv", frames=frames, codec="raw")
    for mode in ("gray", "yuv", "planar"):
        frame = sc.io.Input([video])
        cc = sc.ops.ColorConvert(frame=frame, format=mode)
        out = sp.NamedStream(sc, f"ccv_{mode}")
        sc.run(sc.io.Output(cc, [out]), sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)
        outs = list(sp.NamedVideoStream(sc, f"ccv_{mode}").load())
        f = frames[0].astype(np.float32)
        if mode == "gray":
            assert outs[0].shape == (24, 32, 1)
            ref = (0.299 * f[..., 0] + 0.587 * f[..., 1]
                   + 0.114 * f[..., 2] + 0.5).astype(np.uint8)
            np.testing.assert_allclose(outs[0][..., 0].astype(int),
                                       ref.astype(int), atol=1)
        elif mode == "yuv":
            assert outs[0].shape == (24, 32, 3)
            ref_y = (0.299 * f[..., 0] + 0.587 * f[..., 1]
                     + 0.114 * f[..., 2] + 0.5).astype(np.uint8)
            np.testing.assert_allclose(outs[0][..., 0].astype(int),
                                       ref_y.astype(int), atol=1)
        else:
            assert outs[0].shape == (3, 24, 32)
            np.testing.assert_array_equal(
                outs[0], np.transpose(frames[0], (2, 0, 1)))


def decode_png_filter0(blob):
    """Minimal PNG reader for the encoder's output (filter-0 scanlines)."""
    import struct
    import zlib
    assert blob[:8] == b"\x89PNG\r\n\x1a\x0a"
    pos = 8
    idat = b""
    w = h = ct = None
    while pos < len(blob):
        ln, typ = struct.unpack(">I4s", blob[pos:pos + 8])
        data = blob[pos + 8:pos + 8 + ln]
        if typ == b"IHDR":
            w, h, depth, ct = struct.unpack(">IIBB", data[:10])
            assert depth == 8
        elif typ == b"IDAT":
            idat += data
        pos += 12 + ln
    c = 3 if ct == 2 else 1
    raw = zlib.decompress(idat)
    stride = 1 + w * c
    rows = [raw[i * stride + 1:(i + 1) * stride] for i in range(h)]
    assert all(raw[i * stride] == 0 for i in range(h))
    return np.frombuffer(b"".join(rows), np.uint8).reshape(h, w, c)


def test_image_encoder_png(sc):
    frames = make_video(n=3, h=24, w=32)
    video = sp.NamedVideoStream(sc, "png", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    enc = sc.ops.ImageEncoder(frame=frame, format="png")
    out = sp.NamedStream(sc, "png_out")
    sc.run(sc.io.Output(enc, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 3
    for i, blob in enumerate(rows):
        np.testing.assert_array_equal(decode_png_filter0(blob), frames[i])


def test_compressed_output_column(sc):
    """Sink-side codec compression of a processed frame column (parity:
    reference compressed-output tests py_test.py:730-766)."""
    from conftest import make_smooth_video
    frames = make_smooth_video(n=24, h=48, w=64)
    video = sp.NamedVideoStream(sc, "cmp", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    blur = sc.ops.Blur(frame=frame, kernel_size=3)
    blur.compress_video()
    out = sp.NamedStream(sc, "cmp_out")
    sc.run(sc.io.Output(blur, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    got = np.stack(list(sp.NamedVideoStream(sc, "cmp_out").load()))
    assert got.shape == (24, 48, 64, 3)

    # reference output: run the same graph without compression
    frame = sc.io.Input([video])
    blur2 = sc.ops.Blur(frame=frame, kernel_size=3)
    out2 = sp.NamedStream(sc, "cmp_ref")
    sc.run(sc.io.Output(blur2, [out2]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    ref = np.stack(list(sp.NamedVideoStream(sc, "cmp_ref").load()))
    np.testing.assert_array_equal(got, ref)  # svc is lossless

    # compressed table is smaller than the raw one on smooth content
    import os
    for name, limit in (("cmp_out", 0.8),):
        info = sc.table_info(name)
        tdir = os.path.join(sc._db_path, "tables", str(info["id"]))
        total = sum(os.path.getsize(os.path.join(tdir, f))
                    for f in os.listdir(tdir))
        assert total < frames.nbytes * limit, (total, frames.nbytes)


def test_config_file(tmp_path, monkeypatch):
    cfg = tmp_path / "scanner_amd.toml"
    cfg.write_text(
        '[storage]\ntype = "posix"\ndb_path = "%s"\n' % (tmp_path / "cdb"))
    monkeypatch.setenv("SCANNER_AMD_CONFIG", str(cfg))
    sc2 = sp.Client(start_cluster=False)
    assert sc2._db_path == str(tmp_path / "cdb")
    # ops still work against the configured db
    tab = sc2.new_table("cfg_t", ["col"], [[b"x"]])
    assert sc2.has_table("cfg_t")


def test_batch_load(sc):
    vids = [make_video(n=6, seed=s) for s in range(3)]
    streams = [sp.NamedVideoStream(sc, f"bl{i}", frames=v, codec="raw")
               for i, v in enumerate(vids)]
    loaded = sc.batch_load(streams)
    assert len(loaded) == 3
    for i, rows in enumerate(loaded):
        np.testing.assert_array_equal(np.stack(rows), vids[i])


def test_stencil_clamps_at_slice_boundary(sc):
    """Stencil windows must clamp at slice-group bounds (REPEAT_EDGE,
    reference dag_analysis.cpp:1450-1469): the last row of a group sees
    (last, last), not the first frame of the next group."""
    frames = np.concatenate([make_textured_pair(dx=3, dy=0, seed=7)[0:1]] * 4
                            + [make_textured_pair(dx=3, dy=0, seed=8)[1:2]]
                            * 4)
    # frames 0-3 identical, frames 4-7 identical, 0->4 is a translation
    video = sp.NamedVideoStream(sc, "slb", frames=frames, codec="raw")

    def flow_mags(sliced):
        frame = sc.io.Input([video])
        col = sc.streams.Slice(frame, sc.partitioner.all(4)) if sliced \
            else frame
        flow = sc.ops.OpticalFlow(frame=col)
        stats = sc.ops.FlowStats(flow=flow)
        col_out = sc.streams.Unslice(stats) if sliced else stats
        out = sp.NamedStream(sc, f"slb_out_{sliced}")
        sc.run(sc.io.Output(col_out, [out]), sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)
        return [np.frombuffer(b, np.float32)[0] for b in out.load()]

    unsliced = flow_mags(False)
    sliced = flow_mags(True)
    assert len(unsliced) == len(sliced) == 8
    # row 3: unsliced pairs (3,4) = translation -> mean|u| near 3;
    # sliced clamps to (3,3) within group -> ~0 flow
    assert unsliced[3] > 0.5, unsliced
    assert sliced[3] < 0.2, sliced
    # interior rows agree
    assert abs(unsliced[1] - sliced[1]) < 1e-4


def test_overlapping_slices(sc):
    """Slice with overlapping range partitions: each group processes its
    full range independently (parity: reference overlapping-slice test
    py_test.py:350-406)."""
    n = 12
    tab = sc.new_table("ovl", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    sliced = sc.streams.Slice(col, sc.partitioner.ranges([(0, 8), (4, 12)]))
    inc = sc.ops.TestIncrement(ignore=sliced)
    unsliced = sc.streams.Unslice(inc)
    out = sp.NamedStream(sc, "ovl_out")
    sc.run(sc.io.Output(unsliced, [out]), sp.PerfParams.manual(4, 16),
           cache_mode=sp.CacheMode.Overwrite)
    vals = [int.from_bytes(b, "little") for b in out.load()]
    # group 0: rows 0-7 -> value row + k+1 (k = offset in group);
    # group 1: rows 4-11 -> same pattern; unslice concatenates both groups
    expect = [i + (i % 8) + 1 for i in range(8)] + \
             [4 + i + (i % 8) + 1 for i in range(8)]
    assert vals == expect, vals


def test_batch_ingest_and_export(sc, tmp_path):
    from scanner_amd.storage import ingest_videos
    good = make_video(n=6)
    bad = np.zeros((3, 4), np.uint8)  # wrong rank -> ingest failure
    streams, failures = ingest_videos(
        sc, [("bi_ok", good), ("bi_bad", bad)], codec="svc")
    assert len(streams) == 1 and streams[0].name == "bi_ok"
    assert len(failures) == 1 and failures[0][0] == "bi_bad"

    out = streams[0].save_npy(str(tmp_path / "clip.npy"))
    np.testing.assert_array_equal(np.load(out), good)


def test_multi_instance_stress(sc):
    """Four pipeline instances over a multi-op graph with a stencil op
    (exercises pool-allocator contention, stencil caches, and per-instance
    kernel sets concurrently)."""
    frames = make_video(n=48, h=32, w=40)
    video = sp.NamedVideoStream(sc, "mi", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    blur = sc.ops.Blur(frame=frame, kernel_size=3)
    hist = sc.ops.Histogram(frame=blur)
    flow = sc.ops.OpticalFlow(frame=frame)
    stats = sc.ops.FlowStats(flow=flow)
    out = sp.NamedStream(sc, "mi_out")
    sc.run(sc.io.Output([hist, stats], [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite, pipeline_instances=4)
    info = sc.table_info("mi_out")
    assert info["num_rows"] == 48
    # spot-check: histogram of blurred frame 0 against CPU reference
    import numpy as np
    rows = list(sp.NamedStream(sc, "mi_out", column="histogram").load())
    got = np.frombuffer(rows[0], np.uint32).reshape(3, 256)
    assert got.sum() == 32 * 40 * 3


def test_stencil_after_stride(sc):
    """A stencil op downstream of Stride sees the SAMPLED domain: flow of
    strided frames pairs sampled neighbors (original rows 3k and 3k+3),
    not adjacent original frames (reference: stencils apply in each op's
    own input domain after DomainSampler remapping)."""
    base = make_textured_pair(h=64, w=96, dx=3, dy=0)[0]
    # frames shift right 1px per ORIGINAL frame -> 3px per sampled step
    frames = np.stack([np.roll(base, i, axis=1) for i in range(12)])
    video = sp.NamedVideoStream(sc, "sas", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    sampled = sc.streams.Stride(frame, [3])
    flow = sc.ops.OpticalFlow(frame=sampled)
    stats = sc.ops.FlowStats(flow=flow)
    out = sp.NamedStream(sc, "sas_out")
    sc.run(sc.io.Output(stats, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = [np.frombuffer(b, np.float32) for b in out.load()]
    assert len(rows) == 4  # ceil(12/3)
    # mean |u| of the first window ~ 3 px (sampled-domain neighbors)
    assert 2.0 < rows[0][0] < 4.0, rows[0]


def test_recovery_drops_uncommitted(sc, tmp_path):
    """Startup recovery garbage-collects uncommitted output tables
    (reference: recover_and_init_database master.cpp:1311-1327) while
    committed tables survive."""
    import msgpack
    from scanner_amd import _core

    frames = make_video(n=8)
    video = sp.NamedVideoStream(sc, "rec_in", frames=frames, codec="raw")

    # Build an executor whose prepare() creates the uncommitted output
    # table, then "crash" (drop it without running).
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "rec_out")
    sink = sc.io.Output(hist, [out])
    graph_bytes, jobs, _, _ = sc._assemble(sink)
    ex = _core.LocalExecutor(sc._db, graph_bytes, msgpack.packb(jobs),
                             sp.PerfParams.manual(4, 8).to_dict(1), [])
    ex.prepare(True)
    del ex
    assert sc.has_table("rec_out")
    assert not sc._db.table_committed("rec_out")

    # A fresh client with recovery drops the uncommitted table, keeps the
    # committed input.
    sc2 = sp.Client(db_path=sc._db_path)
    assert sc2.has_table("rec_in")
    assert not sc2.has_table("rec_out")


def test_profiler_io_counters(sc):
    """Profiler counters record IO byte totals (reference: profiler
    increment counters, e.g. io_write column_sink.cpp:198)."""
    frames = make_video(n=8)
    video = sp.NamedVideoStream(sc, "ctr", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "ctr_out")
    prof = sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
                  cache_mode=sp.CacheMode.Overwrite)
    counters = prof.counters()
    assert counters.get("io_read_bytes", 0) > 0
    assert counters.get("io_write_bytes", 0) >= 8 * 3 * 256 * 4
    assert counters.get("rows", 0) == 8


def test_svc_fuzz_roundtrip(sc):
    """Property test: SVC encode/decode is lossless for arbitrary shapes
    (odd dims, tail groups), content types, packet sizes and sparse seek
    patterns."""
    rng = np.random.RandomState(42)
    for trial in range(6):
        h = int(rng.randint(9, 70))
        w = int(rng.randint(9, 70))
        n = int(rng.randint(3, 40))
        kind = trial % 3
        if kind == 0:      # pure noise (incompressible)
            frames = rng.randint(0, 256, size=(n, h, w, 3)).astype(np.uint8)
        elif kind == 1:    # constant-ish (max compression)
            frames = np.full((n, h, w, 3), 17, np.uint8)
            frames += rng.randint(0, 2, size=frames.shape).astype(np.uint8)
        else:              # smooth moving gradient
            yy, xx = np.mgrid[0:h, 0:w]
            frames = np.stack([((xx + yy + 3 * i) % 256).astype(np.uint8)
                               for i in range(n)])
            frames = np.repeat(frames[..., None], 3, axis=-1)
        name = f"fz{trial}"
        io_pkt = int(rng.choice([3, 7, 16, 64]))
        v = sp.NamedVideoStream(sc, name, frames=frames, codec="svc",
                                io_packet_size=io_pkt)
        got = np.stack(list(v.load()))
        np.testing.assert_array_equal(got, frames, err_msg=f"trial {trial}")
        # sparse pattern
        rows = sorted(rng.choice(n, size=min(n, 5), replace=False).tolist())
        got_sparse = list(v.load(rows=rows))
        for k, r in enumerate(rows):
            np.testing.assert_array_equal(got_sparse[k], frames[r])


def test_variadic_op(sc):
    """Variadic-input C++ op: sc.ops.X(inputs=[...]) concatenates any
    number of input columns (reference: variadic_inputs op.h:77)."""
    n = 5
    a = sc.new_table("va", ["col"],
                     [[f"a{i}".encode()] for i in range(n)])
    b = sc.new_table("vb", ["col"],
                     [[f"b{i}".encode()] for i in range(n)])
    c = sc.new_table("vc", ["col"],
                     [[f"c{i}".encode()] for i in range(n)])
    ca = sc.io.Input([a])
    cb = sc.io.Input([b])
    cc = sc.io.Input([c])
    cat = sc.ops.ConcatBytes(inputs=[ca, cb, cc])
    out = sp.NamedStream(sc, "va_out")
    sc.run(sc.io.Output(cat, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert rows == [f"a{i}b{i}c{i}".encode() for i in range(n)]


def test_per_stream_sampling_args(sc):
    """Different sampling args per stream in one bulk job (reference:
    per-stream Job args, py_test.py:339)."""
    vids = [make_video(n=20, seed=s) for s in range(2)]
    streams = [sp.NamedVideoStream(sc, f"ps{i}", frames=v, codec="raw")
               for i, v in enumerate(vids)]
    frame = sc.io.Input(streams)
    # stream 0 takes rows 0..5, stream 1 rows 10..20
    ranged = sc.streams.Range(frame, [(0, 5), (10, 20)])
    hist = sc.ops.Histogram(frame=ranged)
    outs = [sp.NamedStream(sc, f"ps{i}_h") for i in range(2)]
    sc.run(sc.io.Output(hist, outs), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    assert len(list(outs[0].load())) == 5
    rows1 = list(outs[1].load())
    assert len(rows1) == 10
    got = np.frombuffer(rows1[0], np.uint32).reshape(3, 256)
    np.testing.assert_array_equal(got, ref_histogram(vids[1][10]))


def test_gather_over_svc(sc):
    """Gather with duplicate + unsorted-window rows over the codec path
    (keyframe-span decode must serve repeated rows)."""
    frames = make_video(n=40)
    video = sp.NamedVideoStream(sc, "gsv", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    rows_wanted = [2, 2, 17, 17, 39]
    g = sc.streams.Gather(frame, [rows_wanted])
    hist = sc.ops.Histogram(frame=g)
    out = sp.NamedStream(sc, "gsv_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == len(rows_wanted)
    for k, blob in enumerate(rows):
        got = np.frombuffer(blob, np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got,
                                      ref_histogram(frames[rows_wanted[k]]))


def test_streaming_memory_bounded_by_work_packet(sc):
    """Peak engine memory tracks work_packet_size, not io_packet_size
    (streaming packet execution + liveness frees).

    One 200-frame 320x240 task (io_packet = whole stream) must not
    materialize all ~46 MB of decoded frames; the live set is the decode
    chunk + stencil window + the encoded span."""
    from scanner_amd import _core
    n, h, w = 200, 240, 320
    frames = make_smooth_video(n=n, h=h, w=w)
    decoded_bytes = n * h * w * 3
    video = sp.NamedVideoStream(sc, "memb", frames=frames, codec="svc")

    def run(io_packet, name):
        frame = sc.io.Input([video])
        hist = sc.ops.Histogram(frame=frame)
        out = sp.NamedStream(sc, name)
        _core.mem_reset_peak(-1)
        sc.run(sc.io.Output(hist, [out]),
               sp.PerfParams.manual(8, io_packet),
               cache_mode=sp.CacheMode.Overwrite)
        return _core.mem_stats(-1)["peak"], list(out.load())

    peak_one_task, rows_one = run(n, "memb_one")
    peak_small, rows_small = run(25, "memb_small")
    assert rows_one == rows_small
    assert len(rows_one) == n
    # whole-task materialization would hold ~decoded_bytes live; streaming
    # keeps the live set far below it...
    assert peak_one_task < decoded_bytes * 0.5, \
        f"peak {peak_one_task} vs decoded {decoded_bytes}"
    # ...and within a small factor of the many-small-tasks peak
    assert peak_one_task < 3 * peak_small, (peak_one_task, peak_small)


def test_streaming_memory_bounded_with_stencil(sc):
    """Same bound with a stencil op (OpticalFlow, window [0,1]) — the
    stencil window stays live exactly as long as its consumers need it,
    and the big intermediate flow fields are freed as FlowStats consumes
    them."""
    from scanner_amd import _core
    n, h, w = 60, 144, 192
    frames = make_smooth_video(n=n, h=h, w=w)
    # live heavyweights per row: decoded frame (h*w*3) + flow field
    # (h*w*2*4); whole-task materialization would hold n of each
    per_row = h * w * 3 + h * w * 2 * 4
    video = sp.NamedVideoStream(sc, "membs", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    flow = sc.ops.OpticalFlow(frame=frame)  # stencil [0, 1]
    stats = sc.ops.FlowStats(flow=flow)
    out = sp.NamedStream(sc, "membs_out")
    _core.mem_reset_peak(-1)
    sc.run(sc.io.Output(stats, [out]), sp.PerfParams.manual(4, n),
           cache_mode=sp.CacheMode.Overwrite)
    peak = _core.mem_stats(-1)["peak"]
    rows = list(out.load())
    assert len(rows) == n
    assert peak < per_row * n * 0.5, (peak, per_row * n)


def test_table_megafile(tmp_path):
    """Megafile checkpoint (parity: write_table_megafile
    metadata.cpp:441-530): one object holds every committed table's
    descriptor; a fresh Database serves get_table from it with ZERO
    per-table descriptor reads — proven by deleting the individual
    descriptor files before reopening."""
    import os
    sc = sp.Client(db_path=str(tmp_path / "db"))
    for k in range(3):
        sc.new_table(f"mf{k}", ["col"],
                     [[bytes([k, i])] for i in range(4 + k)])
    sc._db.write_megafile()
    assert (tmp_path / "db" / "table_megafile.bin").exists()
    # remove every per-table descriptor: only the megafile knows them now
    removed = 0
    for d in (tmp_path / "db" / "tables").iterdir():
        desc = d / "descriptor.bin"
        if desc.exists():
            desc.unlink()
            removed += 1
    assert removed == 3
    sc2 = sp.Client(db_path=str(tmp_path / "db"))
    for k in range(3):
        rows = list(sp.NamedStream(sc2, f"mf{k}").load())
        assert rows == [bytes([k, i]) for i in range(4 + k)]


def test_gather_unsorted_streaming(sc):
    """Out-of-order Gather ([39, 5, 20, 2]) through the STREAMING packet
    executor: early output packets need late input rows, so the backward
    watermark pass must pull them forward while liveness counts keep every
    element alive until its last (possibly much later) read."""
    frames = make_video(n=40)
    video = sp.NamedVideoStream(sc, "guns", frames=frames, codec="svc")
    wanted = [39, 5, 20, 2, 20]
    frame = sc.io.Input([video])
    g = sc.streams.Gather(frame, [wanted])
    hist = sc.ops.Histogram(frame=g)
    out = sp.NamedStream(sc, "guns_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 3),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == len(wanted)
    for k, blob in enumerate(rows):
        got = np.frombuffer(blob, np.uint32).reshape(3, 256)
        np.testing.assert_array_equal(got, ref_histogram(frames[wanted[k]]))


def test_concurrent_clients_table_creation(tmp_path):
    """4 OS processes create disjoint tables on ONE posix db concurrently:
    the cross-process db lock (ScopedDbLock, metadata.cpp) must serialize
    metadata updates so no creation is lost."""
    import subprocess
    import sys as _sys
    import os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    db = str(tmp_path / "db")
    script = tmp_path / "mk.py"
    script.write_text(
        "import sys\n"
        f"sys.path.insert(0, {repo!r})\n"
        "import scanner_amd as sp\n"
        "idx = int(sys.argv[1])\n"
        f"sc = sp.Client(db_path={db!r}, recover=False)\n"
        "for k in range(6):\n"
        "    sc.new_table(f't{idx}_{k}', ['col'], [[b'12345678']])\n")
    procs = [subprocess.Popen([_sys.executable, str(script), str(i)],
                              stdout=subprocess.PIPE,
                              stderr=subprocess.STDOUT)
             for i in range(4)]
    for p in procs:
        out, _ = p.communicate(timeout=120)
        assert p.returncode == 0, out.decode()
    sc = sp.Client(db_path=db)
    assert len(sc.table_names()) == 24


def test_concurrent_jobs_shared_db(tmp_path):
    """3 OS processes each run a full ingest+histogram job against ONE
    shared posix db concurrently (separate output tables): metadata
    commits serialize under the db lock and every job's results stay
    exact."""
    import subprocess
    import sys as _sys
    import os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    db = str(tmp_path / "db")
    script = tmp_path / "job.py"
    script.write_text(
        "import sys\n"
        f"sys.path.insert(0, {repo!r})\n"
        "import numpy as np\n"
        "import scanner_amd as sp\n"
        "idx = int(sys.argv[1])\n"
        f"sc = sp.Client(db_path={db!r}, recover=False)\n"
        "rng = np.random.RandomState(idx)\n"
        "frames = rng.randint(0, 255, (10, 32, 40, 3)).astype(np.uint8)\n"
        "v = sp.NamedVideoStream(sc, f'v{idx}', frames=frames,\n"
        "                        codec='svc')\n"
        "col = sc.io.Input([v])\n"
        "out = sp.NamedStream(sc, f'h{idx}')\n"
        "sc.run(sc.io.Output(sc.ops.Histogram(frame=col), [out]),\n"
        "       sp.PerfParams.manual(2, 4),\n"
        "       cache_mode=sp.CacheMode.Overwrite)\n"
        "rows = list(out.load())\n"
        "assert len(rows) == 10\n"
        "for i, b in enumerate(rows):\n"
        "    got = np.frombuffer(b, np.uint32).reshape(3, 256)\n"
        "    exp = np.stack([np.bincount(frames[i][:, :, c].ravel(),\n"
        "                                minlength=256) for c in range(3)])\n"
        "    assert (got == exp).all(), f'row {i} mismatch'\n")
    procs = [subprocess.Popen([_sys.executable, str(script), str(i)],
                              stdout=subprocess.PIPE,
                              stderr=subprocess.STDOUT)
             for i in range(3)]
    for p in procs:
        out, _ = p.communicate(timeout=200)
        assert p.returncode == 0, out.decode()[-2000:]


def test_large_row_count_linear(sc):
    """50k rows through plan/liveness/streaming in one job — guards the
    per-row bookkeeping (task plans, read-count maps, liveness frees)
    against accidental quadratic behavior; finishes in well under the
    suite timeout or something regressed."""
    import time
    n = 50000
    tab = sc.new_table("big_lin", ["col"],
                       [[int(i).to_bytes(8, "little")] for i in range(n)])
    col = sc.io.Input([tab])
    out = sp.NamedStream(sc, "big_lin_out")
    t0 = time.time()
    sc.run(sc.io.Output(sc.ops.TestIncrement(ignore=col), [out]),
           sp.PerfParams.manual(64, 512),
           cache_mode=sp.CacheMode.Overwrite)
    elapsed = time.time() - t0
    vals = [int.from_bytes(b, "little") for b in out.load()]
    assert vals == [i + (i % 512) + 1 for i in range(n)]
    assert elapsed < 30, f"50k rows took {elapsed:.1f}s (expected < 1s warm)"
