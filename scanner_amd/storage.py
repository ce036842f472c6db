"""Stored streams (parity: python/scannerpy/storage.py).

NamedStream / NamedVideoStream wrap a table (or a video column of one) in
the scanner database; they are the units sc.io.Input/Output bind per job."""
import numpy as np

from .common import ColumnType, FrameType, ScannerException


class NamedStream:
    """A Bytes column stored as a table (parity: storage.py:250-374)."""

    def __init__(self, sc, name, column=None):
        self._sc = sc
        self.name = name
        self._column = column  # default: first column

    @property
    def is_frame(self):
        return False

    def committed(self):
        return (self._sc._db.has_table(self.name)
                and self._sc._db.table_committed(self.name))

    def exists(self):
        return self._sc._db.has_table(self.name)

    def delete(self):
        self._sc._db.delete_table(self.name)

    def len(self):
        return self._sc._db.table_info(self.name)["num_rows"]

    def column_name(self):
        if self._column:
            return self._column
        info = self._sc._db.table_info(self.name)
        return info["columns"][0][0]

    def load(self, fn=None, rows=None):
        """Yield deserialized rows (None for null elements)."""
        from . import _core
        data = _core.read_column(self._sc._db, self.name, self.column_name(),
                                 list(rows) if rows is not None else [])
        for item in data:
            if item is None:
                yield None
            elif fn is not None:
                yield fn(item)
            else:
                yield item


def ingest_videos(sc, entries, codec="svc", io_packet_size=128,
                  inplace=False):
    """Batch-ingest videos; returns (streams, failures) where failures is
    a list of (name, error) for inputs that could not be ingested (parity:
    Client.ingest_videos + FailedVideo reporting, ingest.cpp:867).

    entries: list of (name, frames-or-npy-path).
    """
    streams, failures = [], []
    for name, src in entries:
        try:
            if isinstance(src, str):
                streams.append(NamedVideoStream(sc, name, path=src,
                                                codec=codec,
                                                io_packet_size=io_packet_size))
            else:
                streams.append(NamedVideoStream(sc, name, frames=src,
                                                codec=codec,
                                                io_packet_size=io_packet_size))
        except Exception as e:
            failures.append((name, f"{type(e).__name__}: {e}"))
    return streams, failures


class NamedVideoStream(NamedStream):
    """A frame column stored as a (possibly codec-compressed) video table
    (parity: NamedVideoStream storage.py:250-374). With no codec libraries
    in this environment, ingest takes raw frames (numpy [N,H,W,C] u8) or an
    .npy path and stores them raw or SVC-compressed."""

    def __init__(self, sc, name, path=None, frames=None, codec="svc",
                 column=None, io_packet_size=128):
        super().__init__(sc, name, column)
        if path is not None and not self.committed() and \
                path.lower().endswith((".mp4", ".h264", ".264")):
            # real video file: demux + H.264 keyframe index (pure parsing;
            # parity: ingest.cpp:175-380). Decode of the result needs the
            # rocDecode/VCN hardware decoder.
            from . import _core
            _core.ingest_video_file(self._sc._db, self.name,
                                    self._column or "frame", path)
            return
        if frames is None and path is not None:
            frames = np.load(path)
        if frames is not None and not self.committed():
            self._ingest(frames, codec, io_packet_size)

    @property
    def is_frame(self):
        return True

    def _ingest(self, frames, codec, io_packet_size):
        from . import _core
        frames = np.asarray(frames)
        if frames.dtype != np.uint8:
            # an implicit astype here silently mangled float frames into
            # u8 garbage; video streams are byte-typed — make the caller
            # quantize explicitly
            raise ScannerException(
                f"video frames must be uint8, got {frames.dtype} "
                "(convert explicitly, e.g. (x*255).clip(0,255)"
                ".astype(np.uint8))")
        frames = np.ascontiguousarray(frames)
        if frames.ndim != 4:
            raise ScannerException("frames must be [N,H,W,C] u8")
        _core.write_video_table(self._sc._db, self.name, self._column or
                                "frame", frames, io_packet_size, codec)

    def load(self, fn=None, rows=None):
        from . import _core
        data = _core.read_column(self._sc._db, self.name, self.column_name(),
                                 list(rows) if rows is not None else [])
        for item in data:
            if item is None:
                yield None
                continue
            buf, shape, ftype = item
            dtype = {0: np.uint8, 1: np.uint16, 2: np.float32,
                     3: np.float64}[ftype]
            arr = np.frombuffer(buf, dtype=dtype).reshape(shape)
            yield fn(arr) if fn is not None else arr


    def save_npy(self, path, rows=None):
        """Export decoded frames to an .npy file (for SVC/raw tables; the
        SVC stream itself is the compressed representation and .npy is the
        decoded interchange export)."""
        frames = np.stack(list(self.load(rows=rows)))
        np.save(path, frames)
        return path

    def save_mp4(self, path, fps=30.0):
        """Remux an ingested H.264 table back into a playable .mp4
        (no transcode, so no codec library needed; parity:
        NamedVideoStream.save_mp4, storage.py:353-374). Only valid for
        tables ingested from real video (codec 'h264')."""
        from . import _core
        _core.export_mp4(self._sc._db, self.name, self.column_name(),
                         path, fps)
        return path
