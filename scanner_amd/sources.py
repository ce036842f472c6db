"""User source/sink DSL: sc.sources.<Name> / sc.sinks.<Name>.

Parity: python/scannerpy source.py / sink.py over the C++ SDK
(csrc/ops/source.h — Source::read / Enumerator::total_elements /
Sink::write + SCA_REGISTER_SOURCE/SINK). The built-in `Files`
source reads one row per file path; the `Files` sink writes each output
row to a file. Custom sources registered by op plugins are reachable by
name the same way."""
from .common import ColumnType, ScannerException
from .op import Op, _as_column


def _per_job(lst):
    """Normalize per-job args: a flat list/scalar means one job."""
    if not isinstance(lst, (list, tuple)) or not lst:
        raise ScannerException("source/sink args must be a non-empty list")
    return list(lst)


class SourcesGenerator:
    def __init__(self, client):
        self._client = client

    def Files(self, paths, column="col"):
        """One row per file. `paths`: list of paths (one job) or list of
        per-job path lists."""
        if isinstance(paths[0], str):
            paths = [paths]
        return self.custom("Files", [{"paths": list(p)} for p in paths],
                           column=column)

    def custom(self, name, args_per_job, column="col", is_frame=False):
        """Bind a registered C++ source (SCA_REGISTER_SOURCE) as the
        graph's Input; `args_per_job` is one msgpack-able dict per job."""
        from . import _core
        if name not in _core.registered_sources():
            raise ScannerException(f"unknown source '{name}' (registered: "
                                   f"{_core.registered_sources()})")
        args_per_job = _per_job(args_per_job)
        op = Op(self._client, "Input", [],
                args={"column": column, "is_frame": is_frame},
                output_columns=[(column, ColumnType.Video if is_frame
                                 else ColumnType.Bytes)])
        op._streams = [None] * len(args_per_job)
        op._source = name
        op._source_args = args_per_job
        return op._single()


class SinksGenerator:
    def __init__(self, client):
        self._client = client

    def Files(self, columns, dirs, ext="bin"):
        """Write each row of each column to <dir>/c<col>_<row>.<ext>.
        `dirs`: one directory (one job) or a per-job list."""
        if isinstance(dirs, str):
            dirs = [dirs]
        return self.custom("Files", columns,
                           [{"dir": d, "ext": ext} for d in dirs])

    def custom(self, name, columns, args_per_job):
        from . import _core
        if name not in _core.registered_sinks():
            raise ScannerException(f"unknown sink '{name}' (registered: "
                                   f"{_core.registered_sinks()})")
        if not isinstance(columns, (list, tuple)):
            columns = [columns]
        cols = [_as_column(c) for c in columns]
        args_per_job = _per_job(args_per_job)
        op = Op(self._client, "Output", cols, args=None, output_columns=[])
        op._streams = [None] * len(args_per_job)
        op._sink = name
        op._sink_args = args_per_job
        return op
