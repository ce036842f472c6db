"""Stream-operation DSL (parity: python/scannerpy/streams.py).

Each method inserts a builtin Sample/Space/Slice/Unslice op whose
per-stream sampling args are carried in the job bindings."""
from .common import ColumnType, ScannerException, SliceList
from .op import Op, OpColumn, _as_column


def _range_args(r):
    if isinstance(r, dict):
        return {"kind": "Range", "starts": [r["start"]], "ends": [r["end"]]}
    return {"kind": "Range", "starts": [r[0]], "ends": [r[1]]}


def _pergroup(x, build):
    if isinstance(x, SliceList):
        return {"kind": "PerGroup", "groups": [build(e) for e in x]}
    return build(x)


def _frame_variant(col, base):
    return base + ("Frame" if col.type == ColumnType.Video else "")


class StreamsGenerator:
    """sc.streams.* (parity: StreamsGenerator streams.py)."""

    def __init__(self, client):
        self._client = client

    @staticmethod
    def _positive(v, what):
        """The engine's samplers clamp stride/spacing to 1 defensively; at
        the API a non-positive value is a caller bug — reject it instead
        of silently reinterpreting (same policy as negative Gather rows)."""
        if v < 1:
            raise ScannerException(f"{what} must be >= 1, got {v}")
        return v

    def _sample(self, col, kind, per_stream_args):
        col = _as_column(col)
        name = _frame_variant(col, "Sample")
        op = Op(self._client, name, [col],
                output_columns=[(col.name, col.type)])
        op._sampling_kind = kind
        op._sampling_args = per_stream_args
        return op._single()

    def _space(self, col, kind, per_stream_args):
        col = _as_column(col)
        name = _frame_variant(col, "Space")
        op = Op(self._client, name, [col],
                output_columns=[(col.name, col.type)])
        op._sampling_kind = kind
        op._sampling_args = per_stream_args
        return op._single()

    def All(self, col):
        return self._sample(col, "All", None)

    def Stride(self, col, strides):
        """strides: int or per-stream list of ints."""
        return self._sample(col, "Strided", [
            {"stride": self._positive(s, "stride")}
            for s in self._bcast(strides)])

    def Range(self, col, ranges):
        """ranges: (start, end) or per-stream list of (start, end) or
        SliceList of {'start','end'} for per-slice-group sampling."""
        return self._sample(col, "Range", [
            _pergroup(r, _range_args)
            for r in self._bcast(ranges, tup=True)])

    def Ranges(self, col, intervals):
        """intervals: per-stream list of [(s0,e0),(s1,e1),...]."""
        return self._sample(col, "StridedRanges", [
            {"stride": 1,
             "starts": [s for s, _ in iv],
             "ends": [e for _, e in iv]}
            for iv in intervals])

    def StridedRange(self, col, ranges):
        """ranges: per-stream list of (start, end, stride)."""
        return self._sample(col, "StridedRange", [
            {"stride": self._positive(r[2], "stride"),
             "starts": [r[0]], "ends": [r[1]]}
            for r in self._bcast(ranges, tup=True)])

    def StridedRanges(self, col, intervals=None, stride=1):
        """intervals: per-stream list of [(s,e),...], one stride."""
        return self._sample(col, "StridedRanges", [
            {"stride": self._positive(stride, "stride"),
             "starts": [s for s, _ in iv],
             "ends": [e for _, e in iv]}
            for iv in intervals])

    def Gather(self, col, rows):
        """rows: per-stream list of row-index lists (each >= 0; unsorted
        and duplicated rows are allowed)."""
        rows = [list(r) for r in rows]
        for r in rows:
            for x in r:
                if x < 0:
                    # -1 is the ENGINE's null-element sentinel (RepeatNull
                    # gaps); letting it through Gather would silently turn
                    # a user indexing bug into null output rows
                    raise ScannerException(
                        f"Gather rows must be >= 0, got {x}")
        return self._sample(col, "Gather", [{"rows": r} for r in rows])

    def Repeat(self, col, spacings):
        return self._space(col, "Repeat", [
            {"spacing": self._positive(s, "spacing")}
            for s in self._bcast(spacings)])

    def RepeatNull(self, col, spacings):
        return self._space(col, "RepeatNull", [
            {"spacing": self._positive(s, "spacing")}
            for s in self._bcast(spacings)])

    def Slice(self, col, partitions):
        """partitions: per-stream partitioner spec from sc.partitioner.*"""
        col = _as_column(col)
        name = _frame_variant(col, "Slice")
        op = Op(self._client, name, [col],
                output_columns=[(col.name, col.type)])
        op._sampling_kind = "partition"
        op._sampling_args = partitions if isinstance(partitions, list) \
            else [partitions]
        return op._single()

    def Unslice(self, col):
        col = _as_column(col)
        name = _frame_variant(col, "Unslice")
        op = Op(self._client, name, [col],
                output_columns=[(col.name, col.type)])
        return op._single()

    @staticmethod
    def _bcast(v, tup=False):
        if tup:
            if isinstance(v, tuple):
                return [v]
            return v
        if isinstance(v, (int, float)):
            return [v]
        return v


class PartitionerGenerator:
    """sc.partitioner.* (parity: python/scannerpy/partitioner.py)."""

    @staticmethod
    def _group(v):
        if int(v) < 1:
            raise ScannerException(f"slice group size must be >= 1, got {v}")
        return int(v)

    def all(self, group_size=None):
        if group_size is None:
            return {"kind": "All"}
        return {"kind": "Strided", "stride": self._group(group_size)}

    def strided(self, group_size):
        return {"kind": "Strided", "stride": self._group(group_size)}

    def ranges(self, intervals):
        return {"kind": "Ranges",
                "starts": [s for s, _ in intervals],
                "ends": [e for _, e in intervals]}

    def strided_ranges(self, intervals, stride=1):
        if stride != 1:
            raise ScannerException(
                "strided slice groups with stride != 1 not supported")
        return self.ranges(intervals)
