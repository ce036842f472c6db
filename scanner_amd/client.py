"""Client: the user entry point (parity: python/scannerpy/client.py).

Local (in-process) execution runs the C++ engine directly; distributed
execution (master/workers over gRPC) goes through scanner_amd.master /
scanner_amd.worker, reusing the same graph assembly done here."""
import os
import tempfile

import msgpack

from . import _core
from .common import (CacheMode, ColumnType, DeviceType, PerfParams,
                     ScannerException)
from .io import IOGenerator
from .op import Op, OpColumn, OpGenerator
from .profiler import Profile
from .storage import NamedStream, NamedVideoStream
from .streams import PartitionerGenerator, StreamsGenerator


class Client:
    def __init__(self, db_path=None, master=None, workers=None,
                 start_cluster=True, recover=None, config=None,
                 config_path=None, storage_type=None, bucket=None):
        if config is None and (config_path is not None
                               or db_path is None or master is None):
            from .config import Config
            config = Config(config_path=config_path, db_path=db_path,
                            master=master)
        if config is not None:
            db_path = db_path or config.db_path
            master = master or config.master_address
            storage_type = storage_type or config.storage_type
            bucket = bucket or config.bucket
        self._db_path = db_path or os.path.join(
            tempfile.gettempdir(), "scanner_amd_db")
        self._storage_type = storage_type or "posix"
        self._bucket = bucket or ""
        if self._storage_type == "posix":
            os.makedirs(self._db_path, exist_ok=True)
        self._db = self._open_db()
        if recover is None:
            # Recovery (uncommitted-table GC) belongs to whoever OWNS the
            # db: a standalone local client, or the MASTER at startup
            # (reference: recover_and_init_database, master.cpp:1311). A
            # client connecting to a cluster must NOT recover — a second
            # client constructed while a job is running would garbage-
            # collect the active job's uncommitted output tables (this
            # exact race flaked the churn stress test: workers then fail
            # with "no table 'o0_0'").
            recover = master is None
        if recover:
            self._db.recover()
        # (metadata reloads after cluster runs go through _open_db so the
        # object-store backend survives them — config.py [storage])
        self.ops = OpGenerator(self)
        self.streams = StreamsGenerator(self)
        self.partitioner = PartitionerGenerator()
        self.io = IOGenerator(self)
        from .sources import SinksGenerator, SourcesGenerator
        self.sources = SourcesGenerator(self)
        self.sinks = SinksGenerator(self)
        self._op_info_cache = {}
        self._last_profilers = None
        # distributed mode
        self._master_addr = master
        self._worker_addrs = workers or []
        self._cluster = None
        if master is not None and start_cluster:
            from .master import ClusterClient
            self._cluster = ClusterClient(self, master, self._worker_addrs)

    # ---- registry ----

    def _op_info(self, name):
        if name not in self._op_info_cache:
            self._op_info_cache[name] = _core.op_info(name)
        return self._op_info_cache[name]

    def has_gpu(self):
        return _core.have_gpu()

    # ---- tables ----

    def table_names(self):
        return self._db.table_names()

    def has_table(self, name):
        return self._db.has_table(name)

    def delete_table(self, name):
        self._db.delete_table(name)

    def table_info(self, name):
        return self._db.table_info(name)

    def new_table(self, name, columns, rows):
        """rows: list (per row) of list (per column) of bytes."""
        per_col = [[r[c] for r in rows] for c in range(len(columns))]
        _core.write_bytes_table(self._db, name, columns, per_col, 128)
        return NamedStream(self, name)

    def ingest_video_file(self, path, name, column="frame"):
        """Ingest a real video file (.mp4 with an AVC track, or an Annex-B
        .h264 stream) into a table: demux + keyframe byte-offset index,
        pure parsing (parity: Client.ingest_videos -> ingest.cpp:175-380).
        Returns {num_frames, width, height, codec}."""
        return _core.ingest_video_file(self._db, name, column, path)

    def sequence(self, name):
        return NamedStream(self, name)

    def _open_db(self):
        if self._storage_type == "posix":
            return _core.Database(self._db_path)
        # object-store backend (S3 semantics emulated over a local bucket
        # dir; config.py [storage] type="s3" bucket="...")
        return _core.Database(self._db_path, self._storage_type,
                              self._bucket)

    def summarize(self):
        lines = []
        for t in sorted(self.table_names()):
            info = self.table_info(t)
            cols = ", ".join(n for n, _ in info["columns"])
            lines.append(f"{t}: {info['num_rows']} rows [{cols}]"
                         f"{'' if info['committed'] else ' (uncommitted)'}")
        return "\n".join(lines)

    # ---- graph assembly ----

    def _toposort(self, sink_op):
        order = []
        seen = {}

        def visit(op):
            if id(op) in seen:
                if seen[id(op)] == 1:
                    raise ScannerException("cycle in op graph")
                return
            seen[id(op)] = 1
            for col in op._inputs:
                visit(col.op)
            seen[id(op)] = 2
            order.append(op)

        visit(sink_op)
        return order

    def _assemble(self, sink_op, n_jobs_hint=None):
        """Returns (graph_bytes, jobs_bytes, out_streams, n_jobs)."""
        ops = self._toposort(sink_op)
        idx = {id(op): i for i, op in enumerate(ops)}

        n_jobs = None
        for op in ops:
            if hasattr(op, "_streams"):
                n = len(op._streams)
                if n_jobs is None:
                    n_jobs = n
                elif n != n_jobs:
                    raise ScannerException(
                        f"stream count mismatch: {n} vs {n_jobs}")
        if n_jobs is None:
            raise ScannerException("graph has no Input op")

        g_ops = []
        for op in ops:
            g_ops.append({
                "name": op._name,
                "inputs": [{"op": idx[id(c.op)], "column": c.name}
                           for c in op._inputs],
                "args": msgpack.packb(op._args) if op._args else b"",
                "device": int(op._device),
                "batch": int(op._batch),
                "stencil": [int(s) for s in op._stencil],
                "warmup": int(op._warmup),
            })
        graph_bytes = msgpack.packb({"ops": g_ops})

        def stream_arg(lst, j):
            if lst is None:
                return None
            if len(lst) == 1:
                return lst[0]
            if j >= len(lst):
                raise ScannerException("fewer per-stream args than jobs")
            return lst[j]

        jobs = []
        out_streams = sink_op._streams
        for j in range(n_jobs):
            sources = {}
            sampling = {}
            op_args = {}
            for i, op in enumerate(ops):
                if op._name == "Input":
                    if getattr(op, "_source", None):
                        sources[str(i)] = {
                            "source": op._source,
                            "args": msgpack.packb(op._source_args[j]),
                        }
                        continue
                    s = op._streams[j]
                    sources[str(i)] = {"table": s.name,
                                       "column": s.column_name()}
                elif hasattr(op, "_sampling_kind"):
                    if op._sampling_kind == "partition":
                        args = dict(stream_arg(op._sampling_args, j))
                    elif op._sampling_kind == "All":
                        args = {"kind": "All"}
                    else:
                        args = dict(stream_arg(op._sampling_args, j))
                        args.setdefault("kind", op._sampling_kind)

                    def norm(a):
                        full = {"kind": a.get("kind", "All"),
                                "stride": a.get("stride", 1),
                                "spacing": a.get("spacing", 1),
                                "starts": a.get("starts", []),
                                "ends": a.get("ends", []),
                                "rows": a.get("rows", [])}
                        if a.get("kind") == "PerGroup":
                            full["groups"] = [norm(g) for g in a["groups"]]
                        return full

                    sampling[str(i)] = norm(args)
                elif op._stream_args is not None:
                    op_args[str(i)] = msgpack.packb(
                        stream_arg(op._stream_args, j))
            job = {
                "sources": sources,
                "sampling": sampling,
                "op_args": op_args,
            }
            if getattr(sink_op, "_sink", None):
                job["sink_table"] = ""
                job["sink_name"] = sink_op._sink
                job["sink_args"] = msgpack.packb(sink_op._sink_args[j])
            else:
                job["sink_table"] = out_streams[j].name
            jobs.append(job)
        return graph_bytes, jobs, out_streams, n_jobs

    # ---- run ----

    def run(self, outputs, perf_params=None, cache_mode=CacheMode.Error,
            gpu_ids=None, pipeline_instances=None, show_progress=False):
        """Execute the graph ending at `outputs` (an Output op) for every
        bound stream (parity: Client.run client.py:1282)."""
        if isinstance(outputs, (list, tuple)):
            if len(outputs) != 1:
                raise ScannerException(
                    "multiple Output sinks per run() not yet supported; "
                    "call run() per sink")
            outputs = outputs[0]
        perf = perf_params or PerfParams.estimate()
        graph_bytes, jobs, out_streams, n_jobs = self._assemble(outputs)

        # An output stream naming a table that is also an INPUT of the
        # same job would be deleted (Overwrite) or half-written before the
        # job reads it — destroying the user's data with a confusing
        # downstream error. Reject before touching storage.
        for j, js in enumerate(jobs):
            s = out_streams[j]
            if s is None:
                continue
            in_tables = {src["table"] for src in js["sources"].values()
                         if "table" in src}
            if s.name in in_tables:
                raise ScannerException(
                    f"output stream '{s.name}' is also an input table of "
                    "the same job; write to a different table")

        # CacheMode handling (parity: client.py:1386-1432)
        keep = []
        for j, js in enumerate(jobs):
            s = out_streams[j]
            if s is None:  # user sink: no output table to cache-check
                keep.append(js)
                continue
            if s.exists() and self._db.table_committed(s.name):
                if cache_mode == CacheMode.Error:
                    raise ScannerException(
                        f"output table '{s.name}' already exists "
                        "(pass cache_mode=CacheMode.Ignore/Overwrite)")
                if cache_mode == CacheMode.Ignore:
                    continue
            keep.append(js)
        if not keep:
            return None
        jobs_bytes = msgpack.packb(keep)

        if self._cluster is not None:
            # ship python ops used by this graph to the master (workers
            # sync them on NewJob)
            from .op import _PY_OP_PICKLES
            for o in msgpack.unpackb(graph_bytes)["ops"]:
                name = o["name"]
                if (name in _PY_OP_PICKLES
                        and name not in self._cluster._py_ops_sent):
                    self._cluster.register_python_op(
                        name, {}, _PY_OP_PICKLES[name])
            result = self._cluster.run_job(graph_bytes, jobs_bytes, perf,
                                           show_progress=show_progress)
            # reload metadata: output tables were created/committed by the
            # master process (shared storage)
            self._db = self._open_db()
            if result is not None:
                self._last_profilers = result._profilers
            return result

        # local execution
        if gpu_ids is None:
            needs_gpu = any(
                o["device"] == int(DeviceType.GPU)
                for o in msgpack.unpackb(graph_bytes)["ops"])
            gpu_ids = list(range(_core.gpu_device_count())) if needs_gpu \
                else []
        if pipeline_instances is None:
            pipeline_instances = (perf.pipeline_instances_per_node
                                  or (len(gpu_ids) if gpu_ids else 1))
        ex = _core.LocalExecutor(self._db, graph_bytes, jobs_bytes,
                                 perf.to_dict(pipeline_instances), gpu_ids)
        if show_progress:
            # run() releases the GIL; poll task completion meanwhile
            # (parity: wait_on_job progress bar, client.py:1188-1261)
            import sys
            import threading
            ex.prepare(True)
            total = len(ex.all_tasks())
            done = threading.Event()

            def _progress():
                while not done.wait(0.5):
                    print(f"\r[scanner] {ex.tasks_done()}/{total} tasks",
                          end="", file=sys.stderr, flush=True)
                print(f"\r[scanner] {ex.tasks_done()}/{total} tasks",
                      file=sys.stderr, flush=True)

            t = threading.Thread(target=_progress, daemon=True)
            t.start()
            try:
                ex.run()
            finally:
                done.set()
                t.join()
        else:
            ex.run()
        self._last_profilers = ex.profilers()
        return Profile(self._last_profilers)

    def load_op(self, path):
        """Load a user C++/HIP op plugin .so built with tools/build_op.py
        (parity: Client.load_op client.py:514 — REGISTER_OP static
        registrars run at load time)."""
        _core.load_op_library(path)
        self._op_info_cache.clear()

    def batch_load(self, streams, fn=None, workers=8, rows=None):
        """Load several streams' rows in parallel (parity: Client.batch_load
        client.py:1270-1281 — thread-pooled column fetch). Returns a list of
        per-stream row lists, in input order."""
        from concurrent.futures import ThreadPoolExecutor

        def one(s):
            return list(s.load(fn=fn, rows=rows))

        with ThreadPoolExecutor(max_workers=workers) as pool:
            return list(pool.map(one, streams))

    def profile(self):
        return Profile(self._last_profilers or [])

    def shutdown(self):
        if self._cluster is not None:
            self._cluster.shutdown()
            self._cluster = None
