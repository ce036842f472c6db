"""Master server: job lifecycle + pull-based task scheduling + fault
tolerance (parity: scanner/engine/master.cpp — NextWork/FinishedWork
protocol, worker pinger with removal after 3 failed pings, task
reassignment, per-stream blacklisting after 3 task failures, commit +
checkpoint semantics, late worker join)."""
import collections
import threading
import time
import uuid

import msgpack

from . import _core
from .common import ScannerException
from .rpc import RpcClient, RpcError, RpcServer

PING_INTERVAL = 2.0
PING_FAILURES = 3
TASK_FAILURES = 3
DEFAULT_TASK_TIMEOUT = 600.0


class _Worker:
    def __init__(self, wid, addr):
        self.id = wid
        self.addr = addr
        self.client = RpcClient(addr)
        self.failed_pings = 0
        self.alive = True


class _BulkJob:
    def __init__(self, jid, graph, jobs, perf, py_ops):
        self.id = jid
        self.graph = graph          # msgpack bytes
        self.jobs = jobs            # list of per-stream dicts
        self.perf = perf            # dict
        self.py_ops = py_ops        # [{name, spec, pickled}]
        self.tasks = []             # [(stream, task, start, end)]
        self.to_assign = collections.deque()
        self.active = {}            # (stream, task) -> (worker_id, t_assign)
        self.done = set()
        self.failures = collections.Counter()
        self.blacklisted = set()    # stream indices
        self.task_errors = {}       # stream -> last worker-reported error
        self.stream_tasks = collections.Counter()
        self.stream_done = collections.Counter()
        self.finalized = set()
        self.complete = threading.Event()
        self.error = None
        self.started = time.time()


class MasterServer:
    def __init__(self, db_path, addr="127.0.0.1:0",
                 no_workers_timeout=30.0, task_timeout=DEFAULT_TASK_TIMEOUT,
                 checkpoint_frequency=1, storage_type="posix", bucket=""):
        self._db_path = db_path
        self._storage = (storage_type or "posix", bucket or "")
        self._db = self._open_db()
        self._db.recover()
        self._workers = {}
        self._next_worker_id = 0
        self._lock = threading.Lock()
        self._job = None            # current bulk job
        self._job_queue = collections.deque()  # queued bulk jobs (FIFO)
        self._jobs_by_id = {}       # id -> _BulkJob (current + queued + done)
        self._job_counter = 0
        self._no_workers_timeout = no_workers_timeout
        self._task_timeout = task_timeout
        self._checkpoint_frequency = max(1, checkpoint_frequency)
        self._jobs_completed = 0
        self._registered_ops = {}   # name -> {spec, pickled}
        self._shutdown = threading.Event()
        self._server = RpcServer(addr, {
            "Ping": self._ping,
            "RegisterWorker": self._register_worker,
            "UnregisterWorker": self._unregister_worker,
            "NewJob": self._new_job,
            "GetJob": self._get_job,
            "NextWork": self._next_work,
            "FinishedWork": self._finished_work,
            "JobStatus": self._job_status,
            "RegisterOp": self._register_op,
            "PokeWatchdog": self._poke,
            "Shutdown": self._shutdown_rpc,
            "CollectProfiles": self._collect_profiles,
        })
        self.port = self._server.port
        self.addr = f"127.0.0.1:{self.port}"
        self._pinger = threading.Thread(target=self._ping_loop, daemon=True)
        self._pinger.start()
        self._monitor = threading.Thread(target=self._monitor_loop,
                                         daemon=True)
        self._monitor.start()

    def _open_db(self):
        # storage config threads to every Database open so a cluster on an
        # object store (kube + S3 semantics) works the same as posix
        st, bucket = self._storage
        if st == "posix":
            return _core.Database(self._db_path)
        return _core.Database(self._db_path, st, bucket)

    # ---- rpc handlers ----

    def _ping(self, req):
        return {"ok": True}

    def _collect_profiles(self, req):
        """Ask every live worker to write its profiler file for a job
        (reference: per-node profile_<node>.bin after a job)."""
        job_id = req["job_id"]
        with self._lock:
            workers = [w for w in self._workers.values() if w.alive]
        written = 0
        for w in workers:
            r = w.client.try_call("WriteProfile", {"job_id": job_id},
                                  timeout=15)
            if r and r.get("written"):
                written += 1
        return {"written": written}

    def _poke(self, req):
        return {"ok": True}

    def _register_worker(self, req):
        with self._lock:
            wid = self._next_worker_id
            self._next_worker_id += 1
            self._workers[wid] = _Worker(wid, req["addr"])
        return {"worker_id": wid}

    def _unregister_worker(self, req):
        with self._lock:
            self._remove_worker(req["worker_id"])
        return {"ok": True}

    def _register_op(self, req):
        self._registered_ops[req["name"]] = {
            "spec": req["spec"], "pickled": req["pickled"]}
        return {"ok": True}

    def _new_job(self, req):
        # id allocation under the lock; the heavy half (graph analysis +
        # output-table creation) runs OUTSIDE it so pings/NextWork from 8+
        # workers are never blocked behind a large job graph (ref: the
        # async master runs process_job on its own thread,
        # master.cpp:1328-1353)
        with self._lock:
            jid = self._job_counter
            self._job_counter += 1
            py_ops = [{"name": n, **v}
                      for n, v in self._registered_ops.items()]
        job = _BulkJob(jid, req["graph"], req["jobs"], req["perf"], py_ops)
        perf_m = dict(job.perf)
        perf_m["cpu_pool_size"] = 0
        perf_m["gpu_pool_size"] = 0
        perf_m["pipeline_instances"] = 1
        # fresh Database handle: the client process may have ingested
        # tables since this one was opened (shared storage); table
        # creation itself is serialized by the db file lock
        db = self._open_db()
        job.ex = _core.LocalExecutor(
            db, job.graph, msgpack.packb(job.jobs), perf_m, [])
        job.ex.prepare(True)
        for (stream, task, start, end) in job.ex.all_tasks():
            t = (stream, task, start, end)
            job.tasks.append(t)
            job.to_assign.append(t)
            job.stream_tasks[stream] += 1
        with self._lock:
            self._db = db
            self._jobs_by_id[jid] = job
            # promote any queued job first so a newly submitted bulk job
            # cannot jump ahead of jobs already waiting in the FIFO queue
            # 
            self._promote_if_done()
            if (self._job is None or self._job.complete.is_set()) \
                    and not self._job_queue:
                self._job = job
                job.started = time.time()
            else:
                # FIFO bulk-job queue (reference: master job_processor
                # queue, master.cpp:1328-1353)
                self._job_queue.append(job)
        return {"job_id": jid, "n_tasks": len(job.tasks)}

    def _get_job(self, req):
        job = self._jobs_by_id.get(req["job_id"])
        if job is None:
            raise ScannerException("no such job")
        return {"job_id": job.id, "graph": job.graph, "jobs": job.jobs,
                "perf": job.perf, "py_ops": job.py_ops}

    def _promote_if_done(self):
        # caller holds lock: activate the next queued bulk job once the
        # current one completes
        if (self._job is not None and self._job.complete.is_set()
                and self._job_queue):
            self._job = self._job_queue.popleft()
            self._job.started = time.time()

    def _next_work(self, req):
        wid = req["worker_id"]
        n = req.get("max_tasks", 1)
        with self._lock:
            self._promote_if_done()
            job = self._job
            if job is None or job.complete.is_set():
                return {"job_id": -1, "tasks": []}
            tasks = []
            while job.to_assign and len(tasks) < n:
                t = job.to_assign.popleft()
                if t[0] in job.blacklisted:
                    continue
                job.active[(t[0], t[1])] = (wid, time.time(), t)
                tasks.append(list(t))
            return {"job_id": job.id, "tasks": tasks,
                    "wait": 0 if tasks else
                    (0.05 if job.active else 0.2)}

    def _finished_work(self, req):
        wid = req["worker_id"]
        stream, task = req["stream"], req["task"]
        with self._lock:
            job = self._job
            if job is None or job.id != req["job_id"]:
                return {"ok": True}
            key = (stream, task)
            entry = job.active.pop(key, None)
            if req.get("success", True):
                if key not in job.done:
                    job.done.add(key)
                    job.stream_done[stream] += 1
                    self._maybe_finalize(job, stream)
            else:
                if req.get("error"):
                    job.task_errors[stream] = req["error"]
                job.failures[key] += 1
                if job.failures[key] >= TASK_FAILURES:
                    # poison stream: blacklist so one bad stream can't sink
                    # the bulk job (reference: blacklist_job master.cpp:2161)
                    job.blacklisted.add(stream)
                    job.active = {k: v for k, v in job.active.items()
                                  if k[0] != stream}
                    self._check_complete(job)
                elif entry is not None:
                    job.to_assign.append(entry[2])
            self._check_complete(job)
        return {"ok": True}

    def _job_status(self, req):
        with self._lock:
            self._promote_if_done()
            job = self._jobs_by_id.get(req.get("job_id", -1))
            if job is None:
                # n_workers always reported (cluster bootstrap polls it)
                return {"exists": False, "n_workers": len(self._workers)}
            return {
                "exists": True,
                "complete": job.complete.is_set(),
                "total_tasks": len(job.tasks),
                "done_tasks": len(job.done),
                "blacklisted_streams": sorted(job.blacklisted),
                "task_errors": {str(k): v
                                for k, v in job.task_errors.items()},
                "error": job.error,
                "n_workers": len(self._workers),
            }

    def _shutdown_rpc(self, req):
        threading.Thread(target=self.shutdown, daemon=True).start()
        return {"ok": True}

    # ---- internals ----

    def _maybe_finalize(self, job, stream):
        # all tasks of this output stream finished -> set end_rows + commit
        # (reference: FinishedWorkHandler table commit master.cpp:1100-1113)
        if (job.stream_done[stream] >= job.stream_tasks[stream]
                and stream not in job.finalized):
            job.finalized.add(stream)
            job.ex.finalize_job(stream)

    def _check_complete(self, job):
        remaining = [t for t in job.tasks
                     if (t[0], t[1]) not in job.done
                     and t[0] not in job.blacklisted]
        if not remaining and not job.complete.is_set():
            job.complete.set()
            # checkpoint: batch committed-table descriptors into the
            # megafile every `checkpoint_frequency` bulk jobs (reference:
            # master.cpp:1109-1112 + write_table_megafile) — off the lock
            self._jobs_completed += 1
            if self._jobs_completed % self._checkpoint_frequency == 0:
                threading.Thread(target=self._write_megafile,
                                 daemon=True).start()

    def _write_megafile(self):
        try:
            self._db.write_megafile()
        except Exception:
            pass  # a failed checkpoint degrades to per-table reads

    def _remove_worker(self, wid):
        # caller holds lock
        w = self._workers.pop(wid, None)
        if w is None:
            return
        w.alive = False
        job = self._job
        if job is not None and not job.complete.is_set():
            # task reassignment (reference: remove_worker master.cpp:2145)
            for key, (owner, _, t) in list(job.active.items()):
                if owner == wid:
                    job.active.pop(key)
                    job.to_assign.append(t)

    def _ping_loop(self):
        while not self._shutdown.is_set():
            time.sleep(PING_INTERVAL)
            with self._lock:
                workers = list(self._workers.values())
            if not workers:
                continue

            # ping all workers concurrently: a dead worker's 5 s timeout
            # must not delay liveness detection of the other 7+ (reference
            # pings each worker on the async completion queue,
            # master.cpp:1837-1965)
            def ping_one(w):
                ok = w.client.try_call("Ping", {}, timeout=5)
                if ok is None:
                    w.failed_pings += 1
                    if w.failed_pings >= PING_FAILURES:
                        with self._lock:
                            self._remove_worker(w.id)
                else:
                    w.failed_pings = 0

            threads = [threading.Thread(target=ping_one, args=(w,),
                                        daemon=True) for w in workers]
            for t in threads:
                t.start()
            for t in threads:
                t.join()

    def _monitor_loop(self):
        while not self._shutdown.is_set():
            time.sleep(0.5)
            with self._lock:
                job = self._job
                if job is None or job.complete.is_set():
                    continue
                now = time.time()
                # straggler/task timeout
                for key, (owner, t0, t) in list(job.active.items()):
                    if now - t0 > self._task_timeout:
                        job.failures[key] += 1
                        job.active.pop(key)
                        if job.failures[key] >= TASK_FAILURES:
                            job.blacklisted.add(t[0])
                        else:
                            job.to_assign.append(t)
                # no workers at all
                if (not self._workers
                        and now - job.started > self._no_workers_timeout):
                    job.error = ("no workers available after "
                                 f"{self._no_workers_timeout}s")
                    job.complete.set()
                self._check_complete(job)

    def shutdown(self):
        self._shutdown.set()
        with self._lock:
            workers = list(self._workers.values())
        for w in workers:
            w.client.try_call("Shutdown", {}, timeout=5)
        self._server.stop()


def start_master(db_path, addr="127.0.0.1:0", block=False, **kw):
    m = MasterServer(db_path, addr, **kw)
    if block:
        m._server.wait()
    return m


class ClusterClient:
    """Client-side handle for distributed runs (lives inside Client)."""

    def __init__(self, client, master_addr, worker_addrs):
        self._client = client
        self._rpc = RpcClient(master_addr)
        try:
            self._rpc.call("Ping", {}, timeout=10)
        except RpcError as e:
            raise ScannerException(f"cannot reach master: {e}")
        self._heartbeat_stop = threading.Event()
        self._hb = threading.Thread(target=self._heartbeat, daemon=True)
        self._hb.start()
        self._py_ops_sent = set()

    def _heartbeat(self):
        while not self._heartbeat_stop.wait(5.0):
            self._rpc.try_call("PokeWatchdog", {}, timeout=5)

    def register_python_op(self, name, spec, pickled):
        self._rpc.call("RegisterOp",
                       {"name": name, "spec": spec, "pickled": pickled})
        self._py_ops_sent.add(name)

    def run_job(self, graph_bytes, jobs_bytes, perf, show_progress=False):
        jobs = msgpack.unpackb(jobs_bytes, raw=False)
        resp = self._rpc.call("NewJob", {
            "graph": graph_bytes, "jobs": jobs,
            "perf": perf.to_dict(1)}, timeout=120)
        jid = resp["job_id"]
        while True:
            st = self._rpc.call("JobStatus", {"job_id": jid}, timeout=30)
            if not st["exists"]:
                raise ScannerException("job vanished")
            if show_progress:
                print(f"\r[job {jid}] {st['done_tasks']}/{st['total_tasks']}"
                      f" tasks, {st['n_workers']} workers", end="",
                      flush=True)
            if st["complete"]:
                if show_progress:
                    print()
                if st["error"]:
                    raise ScannerException(st["error"])
                if st["blacklisted_streams"]:
                    # include the last worker-reported error per stream so
                    # the root cause is in the exception, not lost in a
                    # worker subprocess log
                    errs = st.get("task_errors", {})
                    raise ScannerException(
                        "streams failed (blacklisted after repeated task "
                        f"failures): {st['blacklisted_streams']}"
                        + (f"; last errors: {errs}" if errs else ""))
                return self._load_profiles(jid)
            time.sleep(0.1)

    def _load_profiles(self, jid):
        """Gather per-worker profiler files written via CollectProfiles
        (parity: Profile parsing jobs/<id>/profile_<node>.bin)."""
        import glob
        import os

        from .profiler import Profile
        self._rpc.try_call("CollectProfiles", {"job_id": jid}, timeout=30)
        profs = []
        pat = os.path.join(self._client._db_path, "jobs", str(jid),
                           "profile_*.bin")
        for path in sorted(glob.glob(pat)):
            with open(path, "rb") as f:
                profs.extend(msgpack.unpackb(f.read(), raw=False))
        return Profile(profs)

    def shutdown(self):
        self._heartbeat_stop.set()
        self._rpc.try_call("Shutdown", {}, timeout=5)


def main():
    import argparse
    ap = argparse.ArgumentParser(
        description="scanner_amd master server (reference: start_master)")
    ap.add_argument("--db-path", required=True)
    ap.add_argument("--addr", default="0.0.0.0:5001")
    ap.add_argument("--no-workers-timeout", type=float, default=30.0)
    ap.add_argument("--task-timeout", type=float, default=600.0)
    ap.add_argument("--storage-type", default="posix",
                    help="posix | s3 (object store; see config.py)")
    ap.add_argument("--bucket", default="")
    args = ap.parse_args()
    m = MasterServer(args.db_path, args.addr,
                     no_workers_timeout=args.no_workers_timeout,
                     task_timeout=args.task_timeout,
                     storage_type=args.storage_type, bucket=args.bucket)
    print(f"master listening on {m.addr} (db: {args.db_path})", flush=True)
    m._server.wait()


if __name__ == "__main__":
    main()
