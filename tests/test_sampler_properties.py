"""Property-based sampler/space/slice tests (hypothesis).

The reference's samplers are covered by example-based tests only
(tests/py_test.py test_stride/test_range/...); here the whole sampling-arg
space is swept against an independent Python model of the row mapping
(reference semantics: DomainSamplers sampler.cpp:33-463). Each example
runs a REAL end-to-end job (table -> Sample/Space -> Output) and compares
the produced rows — value by value, None for null elements — to the
model. Fixed derandomized profile so CI runs are reproducible.
"""
import os
import sys
import tempfile
from typing import Sequence

from hypothesis import HealthCheck, given, settings, strategies as st

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import scanner_amd as sp  # noqa: E402

N = 23  # upstream rows in the shared input table

_client = None
_table = None


def client():
    global _client, _table
    if _client is None:
        tmp = tempfile.mkdtemp(prefix="scanner_prop_")
        _client = sp.Client(db_path=os.path.join(tmp, "db"))
        _table = _client.new_table(
            "prop_in", ["col"],
            [[int(i).to_bytes(8, "little")] for i in range(N)])
    return _client, _table


# ---- strategies over valid sampling specs ----

def intervals_strategy(n):
    """Sorted non-overlapping non-empty [start, end) intervals within n."""
    return st.lists(st.integers(0, n), min_size=2, max_size=6).map(
        lambda cuts: [(a, b) for a, b in
                      zip(*(iter(sorted(set(cuts))),) * 2) if a < b]
    ).filter(lambda iv: len(iv) >= 1)


spec = st.one_of(
    st.tuples(st.just("stride"), st.integers(1, 9)),
    st.tuples(st.just("range"),
              st.tuples(st.integers(0, N - 1), st.integers(0, N)).map(
                  lambda ab: (min(ab), max(ab)) if ab[0] != ab[1]
                  else (ab[0], ab[0] + 1))),
    st.tuples(st.just("ranges"), intervals_strategy(N)),
    st.tuples(st.just("strided_range"),
              st.tuples(st.integers(0, N - 1), st.integers(1, N),
                        st.integers(1, 6)).map(
                  lambda abs_: (min(abs_[0], abs_[1] - 1), abs_[1],
                                abs_[2]) if abs_[0] < abs_[1]
                  else (abs_[1] - 1, abs_[1], abs_[2]))),
    st.tuples(st.just("strided_ranges"),
              st.tuples(intervals_strategy(N), st.integers(1, 5))),
    st.tuples(st.just("gather"),
              st.lists(st.integers(0, N - 1), min_size=1, max_size=30)),
    st.tuples(st.just("repeat"), st.integers(1, 4)),
    st.tuples(st.just("repeat_null"), st.integers(1, 4)),
)


def model_rows(kind, args, n):
    """Upstream row index per downstream row (None = null element)."""
    if kind == "stride":
        return list(range(0, n, args))
    if kind == "range":
        a, b = args
        return list(range(a, b))
    if kind == "ranges":
        return [i for a, b in args for i in range(a, b)]
    if kind == "strided_range":
        a, b, s = args
        return list(range(a, b, s))
    if kind == "strided_ranges":
        iv, s = args
        return [i for a, b in iv for i in range(a, b, s)]
    if kind == "gather":
        return list(args)
    if kind == "repeat":
        return [i // args for i in range(n * args)]
    if kind == "repeat_null":
        return [i // args if i % args == 0 else None
                for i in range(n * args)]
    raise AssertionError(kind)


def apply_spec(sc, col, kind, args):
    s = sc.streams
    return {
        "stride": lambda: s.Stride(col, [args]),
        "range": lambda: s.Range(col, [args]),
        "ranges": lambda: s.Ranges(col, [args]),
        "strided_range": lambda: s.StridedRange(col, [args]),
        "strided_ranges": lambda: s.StridedRanges(col, [args[0]],
                                                  stride=args[1]),
        "gather": lambda: s.Gather(col, [args]),
        "repeat": lambda: s.Repeat(col, [args]),
        "repeat_null": lambda: s.RepeatNull(col, [args]),
    }[kind]()


@settings(max_examples=30, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(spec=spec)
def test_sampler_matches_model(spec):
    kind, args = spec
    sc, tab = client()
    col = sc.io.Input([tab])
    sampled = apply_spec(sc, col, kind, args)
    out = sp.NamedStream(sc, "prop_out")
    sc.run(sc.io.Output(sampled, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    got = [None if b is None else int.from_bytes(b, "little")
           for b in out.load()]
    assert got == model_rows(kind, args, N), (kind, args)


@sp.register_python_op(name="PropDouble")
def PropDouble(col: bytes) -> bytes:
    return (2 * int.from_bytes(col, "little")).to_bytes(8, "little")


@sp.register_python_op(name="PropStencilSum", stencil=[-1, 0, 1])
def PropStencilSum(cols: Sequence[bytes]) -> bytes:
    s = sum(int.from_bytes(c, "little") for c in cols)
    return s.to_bytes(8, "little")


@st.composite
def pipeline_spec(draw):
    """1-3 stages over a running stream length; samplers change the
    length, ops keep it."""
    stages = []
    length = N
    for _ in range(draw(st.integers(1, 3))):
        kind = draw(st.sampled_from(
            ["stride", "range", "gather", "double", "stencilsum"]))
        if kind == "stride":
            stages.append(("stride", draw(st.integers(1, 5))))
            length = len(range(0, length, stages[-1][1]))
        elif kind == "range":
            a = draw(st.integers(0, length - 1))
            b = draw(st.integers(a + 1, length))
            stages.append(("range", (a, b)))
            length = b - a
        elif kind == "gather":
            rows = draw(st.lists(st.integers(0, length - 1),
                                 min_size=1, max_size=15))
            stages.append(("gather", rows))
            length = len(rows)
        else:
            stages.append((kind, None))
    return stages


@settings(max_examples=25, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(spec=pipeline_spec())
def test_composed_pipeline_matches_model(spec):
    """Chains of samplers and (stateless / stencil) ops: the sampled row
    spaces compose, stencil windows apply in the RESAMPLED space with
    REPEAT_EDGE clamping at its bounds (reference semantics:
    derive_stencil_requirements walks remaps then stencils,
    dag_analysis.cpp:1328+)."""
    sc, tab = client()
    col = sc.io.Input([tab])
    values = list(range(N))  # model: value per row at this stage
    for kind, args in spec:
        if kind == "stride":
            col = sc.streams.Stride(col, [args])
            values = values[::args]
        elif kind == "range":
            col = sc.streams.Range(col, [args])
            values = values[args[0]:args[1]]
        elif kind == "gather":
            col = sc.streams.Gather(col, [args])
            values = [values[r] for r in args]
        elif kind == "double":
            col = sc.ops.PropDouble(col=col)
            values = [2 * v for v in values]
        elif kind == "stencilsum":
            col = sc.ops.PropStencilSum(cols=col)
            m = len(values)
            values = [values[max(i - 1, 0)] + values[i] +
                      values[min(i + 1, m - 1)] for i in range(m)]
    out = sp.NamedStream(sc, "prop_chain_out")
    sc.run(sc.io.Output(col, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    got = [int.from_bytes(b, "little") for b in out.load()]
    assert got == values, spec


def _make_counter(name, warmup):
    @sp.register_python_op(name=name, bounded_state=True, warmup=warmup)
    class _Ctr(sp.Kernel):
        def __init__(self, config, **kw):
            self.c = 0

        def reset(self):
            self.c = 0

        def execute(self, col: bytes) -> bytes:
            self.c += 1
            return (int.from_bytes(col, "little") +
                    self.c).to_bytes(8, "little")


for _w in (0, 1, 2, 3):
    _make_counter(f"PropCtrW{_w}", _w)


@settings(max_examples=25, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(warmup=st.integers(0, 3), io=st.integers(1, 8),
       work=st.integers(1, 8), n=st.integers(1, N))
def test_bounded_state_warmup_matches_model(warmup, io, work, n):
    """Bounded-state warmup across task boundaries for every (warmup,
    io_packet, work_packet, stream length): at each task start the kernel
    resets and re-feeds min(warmup, start) rows of history, so the
    counter's contribution to row r is r - max(0, task_start - warmup) + 1
    (reference: derive_stencil_requirements warmup handling,
    dag_analysis.cpp:1328+; verified to match the engine for the pinned
    example in this file's history)."""
    work = min(work, io)
    sc, _ = client()
    tab_name = f"prop_bs_{n}"
    try:
        tab = sc.new_table(
            tab_name, ["col"],
            [[int(i).to_bytes(8, "little")] for i in range(n)])
    except Exception:
        from scanner_amd.storage import NamedStream as _NS
        tab = _NS(sc, tab_name)  # already created by an earlier example
    col = sc.io.Input([tab])
    ctr = getattr(sc.ops, f"PropCtrW{warmup}")(col=col)
    out = sp.NamedStream(sc, "prop_bs_out")
    sc.run(sc.io.Output(ctr, [out]), sp.PerfParams.manual(work, io),
           cache_mode=sp.CacheMode.Overwrite)
    got = [int.from_bytes(b, "little") for b in out.load()]
    expect = [r + (r - max(0, (r // io) * io - warmup) + 1)
              for r in range(n)]
    assert got == expect, (warmup, io, work, n)


@settings(max_examples=15, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.too_slow])
@given(group_size=st.integers(1, N + 4))
def test_slice_unslice_identity(group_size):
    """Slice into fixed-size groups then Unslice == identity, for every
    group size incl. non-dividing and larger-than-stream."""
    sc, tab = client()
    col = sc.io.Input([tab])
    sliced = sc.streams.Slice(col, sc.partitioner.strided(group_size))
    joined = sc.streams.Unslice(sliced)
    out = sp.NamedStream(sc, "prop_slice_out")
    sc.run(sc.io.Output(joined, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    got = [int.from_bytes(b, "little") for b in out.load()]
    assert got == list(range(N)), group_size
