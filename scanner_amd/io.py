"""Source/sink DSL: sc.io.Input / sc.io.Output (parity:
python/scannerpy/io.py + source.py + sink.py)."""
from .common import ColumnType, ScannerException
from .op import Op, OpColumn, _as_column


class IOGenerator:
    def __init__(self, client):
        self._client = client

    def Input(self, streams):
        """streams: list of NamedStream/NamedVideoStream, one per job."""
        if not isinstance(streams, (list, tuple)) or not streams:
            raise ScannerException("Input takes a non-empty list of streams")
        is_frame = streams[0].is_frame
        for s in streams:
            if s.is_frame != is_frame:
                raise ScannerException("Input streams must be homogeneous")
        col = "frame" if is_frame else "col"
        op = Op(self._client, "Input", [],
                args={"column": col, "is_frame": is_frame},
                output_columns=[(col,
                                 ColumnType.Video if is_frame
                                 else ColumnType.Bytes)])
        op._streams = list(streams)
        return op._single()

    def Output(self, columns, streams):
        """columns: one or a list of op output columns; streams: output
        NamedStreams, one per job."""
        if not isinstance(columns, (list, tuple)):
            columns = [columns]
        cols = [_as_column(c) for c in columns]
        seen = set()
        for c in cols:
            if c.name in seen:
                raise ScannerException(
                    f"duplicate output column name '{c.name}'")
            seen.add(c.name)
        # sink-side codec annotations (OpColumn.compress_video / lossless)
        compress = {c.name: c.op._compress[c.name].get("codec", "svc")
                    for c in cols if c.name in c.op._compress}
        op = Op(self._client, "Output", cols,
                args={"compress": compress} if compress else None,
                output_columns=[])
        op._streams = list(streams)
        return op
