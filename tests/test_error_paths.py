"""Negative-path coverage: invalid DAGs, bad sampling args, missing
tables, kernel input validation — every misuse must raise a clean Python
exception (reference: graph validation dag_analysis.cpp:43 + client-side
checks in scannerpy client.py run()); silent wrong output or a crash is
the failure mode these tests guard against."""
import pytest

import scanner_amd as sp


def int_table(sc, name, n, width=8):
    return sc.new_table(name, ["col"],
                        [[int(i).to_bytes(width, "little")]
                         for i in range(n)])


def test_unknown_op(sc):
    tab = int_table(sc, "ep1", 4)
    with pytest.raises(Exception, match="unknown op"):
        sc.ops.NoSuchOp(col=sc.io.Input([tab]))


def test_missing_input_column(sc):
    tab = int_table(sc, "ep2", 4)
    with pytest.raises(sp.ScannerException, match="missing input column"):
        sc.ops.TestIncrement(bogus=sc.io.Input([tab]))


def test_out_of_domain_range(sc):
    tab = int_table(sc, "ep3", 10)
    col = sc.io.Input([tab])
    r = sc.streams.Range(col, [(5, 100)])
    out = sp.NamedStream(sc, "ep3_out")
    with pytest.raises(Exception, match="out of op domain"):
        sc.run(sc.io.Output(r, [out]), sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)


def test_negative_gather_rejected(sc):
    tab = int_table(sc, "ep4", 10)
    col = sc.io.Input([tab])
    with pytest.raises(sp.ScannerException, match="Gather rows"):
        sc.streams.Gather(col, [[-1, 2]])


def test_load_missing_table(sc):
    with pytest.raises(Exception, match="no table"):
        list(sp.NamedStream(sc, "ep_nope").load())


def test_stream_count_mismatch(sc):
    a = int_table(sc, "ep5a", 4)
    b = int_table(sc, "ep5b", 4)
    col = sc.io.Input([a, b])
    out = sp.NamedStream(sc, "ep5_out")  # 1 output for 2 input streams
    with pytest.raises(sp.ScannerException, match="stream count mismatch"):
        sc.run(sc.io.Output(sc.ops.TestIncrement(ignore=col), [out]),
               sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)


def test_kernel_input_validation_fails_job(sc):
    """A kernel's own input check (TestIncrement wants 8-byte rows)
    surfaces as a job failure with the kernel's message, not a crash."""
    tab = int_table(sc, "ep6", 4, width=2)
    col = sc.io.Input([tab])
    out = sp.NamedStream(sc, "ep6_out")
    with pytest.raises(Exception, match="i64 input"):
        sc.run(sc.io.Output(sc.ops.TestIncrement(ignore=col), [out]),
               sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)


def test_nonpositive_sampling_args_rejected(sc):
    tab = int_table(sc, "ep8", 10)
    for make in (
        lambda c: sc.streams.Stride(c, [0]),
        lambda c: sc.streams.Stride(c, [-2]),
        lambda c: sc.streams.StridedRange(c, [(0, 5, 0)]),
        lambda c: sc.streams.StridedRanges(c, [[(0, 5)]], stride=0),
        lambda c: sc.streams.Repeat(c, [0]),
        lambda c: sc.streams.RepeatNull(c, [-1]),
        lambda c: sc.streams.Slice(c, sc.partitioner.strided(0)),
    ):
        with pytest.raises(sp.ScannerException, match=">= 1"):
            make(sc.io.Input([tab]))


def test_perf_params_validation():
    with pytest.raises(sp.ScannerException, match=">= 1"):
        sp.PerfParams.manual(0, 0)
    with pytest.raises(sp.ScannerException, match="io_packet_size"):
        sp.PerfParams.manual(8, 4)


def test_output_table_is_input_rejected(sc):
    """Writing a job's output over one of its own input tables would
    destroy the input before it is read (Overwrite deletes first) — must
    be rejected before any storage is touched, with the input intact."""
    tab = int_table(sc, "ep9", 10)
    col = sc.io.Input([tab])
    out = sp.NamedStream(sc, "ep9")  # same table!
    with pytest.raises(sp.ScannerException, match="also an input"):
        sc.run(sc.io.Output(sc.ops.TestIncrement(ignore=col), [out]),
               sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)
    assert len(list(sp.NamedStream(sc, "ep9").load())) == 10  # intact


def test_float_frames_rejected(sc):
    """Implicit astype used to mangle float frames into u8 garbage
    silently; now an explicit error."""
    import numpy as np
    with pytest.raises(sp.ScannerException, match="uint8"):
        sp.NamedVideoStream(sc, "epf",
                            frames=np.zeros((2, 8, 8, 3), np.float32),
                            codec="raw")


def test_python_op_cannot_shadow_builtin():
    with pytest.raises(sp.ScannerException, match="already a registered"):
        @sp.register_python_op(name="Histogram")
        def Histogram(col: bytes) -> bytes:  # pragma: no cover
            return col


def test_python_op_exception_message_preserved(sc):
    """A python kernel's exception surfaces in the job failure with its
    type, message, and traceback line."""
    @sp.register_python_op(name="BoomEp")
    def BoomEp(col: bytes) -> bytes:
        raise ValueError("domain-specific detail 12345")

    tab = int_table(sc, "ep10", 4)
    col = sc.io.Input([tab])
    out = sp.NamedStream(sc, "ep10_out")
    with pytest.raises(Exception, match="domain-specific detail 12345"):
        sc.run(sc.io.Output(sc.ops.BoomEp(col=col), [out]),
               sp.PerfParams.manual(4, 8),
               cache_mode=sp.CacheMode.Overwrite)


def frame_table(sc, name):
    import numpy as np
    frames = np.zeros((4, 32, 40, 3), np.uint8)
    return sp.NamedVideoStream(sc, name, frames=frames, codec="raw")


def test_blur_bad_kernel_size_rejected(sc):
    """kernel_size < 1 made the tap radius negative -> division by zero
    (UB, observed as a WEDGED pipeline instance pre-fix)."""
    v = frame_table(sc, "ep11")
    col = sc.io.Input([v])
    out = sp.NamedStream(sc, "ep11_out")
    with pytest.raises(Exception, match="kernel_size must be >= 1"):
        sc.run(sc.io.Output(sc.ops.Blur(frame=col, kernel_size=-3), [out]),
               sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)


def test_crop_negative_offset_rejected(sc):
    """Negative x/y passed the (x + w <= frame_w) bound and read before
    the source buffer (silent OOB read pre-fix)."""
    v = frame_table(sc, "ep12")
    col = sc.io.Input([v])
    out = sp.NamedStream(sc, "ep12_out")
    with pytest.raises(Exception, match="x/y must be >= 0"):
        sc.run(sc.io.Output(
            sc.ops.Crop(frame=col, x=-2, y=0, width=10, height=10), [out]),
            sp.PerfParams.manual(2, 4), cache_mode=sp.CacheMode.Overwrite)


def test_zero_row_job(sc):
    """An empty sampling result is a valid job: zero output rows, no
    error (the engine must handle tasks with nothing to produce)."""
    tab = int_table(sc, "ep7", 10)
    col = sc.io.Input([tab])
    r = sc.streams.Range(col, [(3, 3)])
    out = sp.NamedStream(sc, "ep7_out")
    sc.run(sc.io.Output(r, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    assert list(out.load()) == []


def test_colorconvert_unknown_format(sc):
    v = frame_table(sc, "ep13")
    col = sc.io.Input([v])
    out = sp.NamedStream(sc, "ep13_out")
    with pytest.raises(Exception, match="gray"):
        sc.run(sc.io.Output(
            sc.ops.ColorConvert(frame=col, format="bogus"), [out]),
            sp.PerfParams.manual(2, 4), cache_mode=sp.CacheMode.Overwrite)


def test_resize_bad_dims(sc):
    v = frame_table(sc, "ep14")
    col = sc.io.Input([v])
    out = sp.NamedStream(sc, "ep14_out")
    for w, h in ((-5, 10), (0, 0)):
        with pytest.raises(Exception, match="width/height"):
            sc.run(sc.io.Output(
                sc.ops.Resize(frame=col, width=w, height=h), [out]),
                sp.PerfParams.manual(2, 4),
                cache_mode=sp.CacheMode.Overwrite)


def test_histogram_on_bytes_rejected(sc):
    tab = int_table(sc, "ep15", 4)
    col = sc.io.Input([tab])
    out = sp.NamedStream(sc, "ep15_out")
    with pytest.raises(Exception, match="frame input"):
        sc.run(sc.io.Output(sc.ops.Histogram(frame=col), [out]),
               sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
