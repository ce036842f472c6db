"""Worker server: pulls tasks from the master, runs them on the C++
engine's pipeline instances, acks completions (parity:
scanner/engine/worker.cpp — registration, op sync, pull loop with backoff,
least-loaded dispatch, watchdog self-termination)."""
import argparse
import os
import queue
import threading
import time

import msgpack

from . import _core
from .rpc import RpcClient, RpcError, RpcServer

WATCHDOG_TIMEOUT = 60.0


class WorkerServer:
    def __init__(self, master_addr, db_path, addr="127.0.0.1:0",
                 pipeline_instances=1, gpu_ids=None, watchdog=True,
                 storage_type="posix", bucket=""):
        self._db_path = db_path
        self._storage = (storage_type or "posix", bucket or "")
        self._db = self._open_db()
        self._master = RpcClient(master_addr)
        self._instances = pipeline_instances
        self._gpu_ids = gpu_ids or []
        self._last_poke = time.time()
        self._shutdown = threading.Event()
        self._job_lock = threading.Lock()
        self._job_id = -1
        self._executor = None
        self._registered_py_ops = set()

        self._server = RpcServer(addr, {
            "Ping": self._ping,
            "Shutdown": self._shutdown_rpc,
            "PokeWatchdog": self._poke,
            "WriteProfile": self._write_profile,
        })
        self.port = self._server.port
        self.addr = f"127.0.0.1:{self.port}"
        resp = self._master.call("RegisterWorker", {"addr": self.addr})
        self.worker_id = resp["worker_id"]

        self._task_q = queue.Queue()
        # load-worker stage (parity: dedicated load threads): plans +
        # pinned source reads happen ahead of the pipeline instances
        self._prep_q = queue.Queue(maxsize=pipeline_instances + 2)
        self._threads = []
        for i in range(max(2, pipeline_instances)):
            t = threading.Thread(target=self._loader_loop, daemon=True)
            t.start()
            self._threads.append(t)
        for i in range(pipeline_instances):
            t = threading.Thread(target=self._instance_loop, args=(i,),
                                 daemon=True)
            t.start()
            self._threads.append(t)
        self._pull = threading.Thread(target=self._pull_loop, daemon=True)
        self._pull.start()
        if watchdog:
            self._wd = threading.Thread(target=self._watchdog, daemon=True)
            self._wd.start()

    # ---- rpc ----

    def _ping(self, req):
        self._last_poke = time.time()
        return {"ok": True}

    def _open_db(self):
        st, bucket = self._storage
        if st == "posix":
            return _core.Database(self._db_path)
        return _core.Database(self._db_path, st, bucket)

    def _poke(self, req):
        self._last_poke = time.time()
        return {"ok": True}

    def _shutdown_rpc(self, req):
        threading.Thread(target=self.shutdown, daemon=True).start()
        return {"ok": True}

    def _write_profile(self, req):
        """Serialize this worker's pipeline profilers for the given job to
        shared storage (parity: write_profiler_to_file profiler.h:82 —
        jobs/<id>/profile_<node>.bin, parsed by scannerpy Profile)."""
        job_id = req["job_id"]
        with self._job_lock:
            if self._job_id != job_id or self._executor is None:
                return {"written": False}
            profs = self._executor.profilers()
        d = os.path.join(self._db_path, "jobs", str(job_id))
        os.makedirs(d, exist_ok=True)
        path = os.path.join(d, f"profile_{self.worker_id}.bin")
        with open(path, "wb") as f:
            f.write(msgpack.packb(profs))
        return {"written": True}

    # ---- job setup ----

    def _ensure_job(self, job_id):
        with self._job_lock:
            if self._job_id == job_id and self._executor is not None:
                return
            params = self._master.call("GetJob", {"job_id": job_id},
                                       timeout=60)
            # sync python ops from the master (reference: worker op sync
            # worker.cpp:868-938)
            for op in params.get("py_ops", []):
                if op["name"] in self._registered_py_ops:
                    continue
                import cloudpickle
                reg = cloudpickle.loads(op["pickled"])
                reg()  # re-runs the registration in this process
                self._registered_py_ops.add(op["name"])
            perf = dict(params["perf"])
            perf["pipeline_instances"] = self._instances
            # reload metadata (ingests + master-created output tables)
            self._db = self._open_db()
            ex = _core.LocalExecutor(
                self._db, params["graph"], msgpack.packb(params["jobs"]),
                perf, self._gpu_ids)
            ex.prepare(False)
            self._executor = ex
            self._job_id = job_id

    # ---- loops ----

    def _pull_loop(self):
        backoff = 0.05
        while not self._shutdown.is_set():
            if self._task_q.qsize() >= self._instances:
                time.sleep(0.01)
                continue
            try:
                resp = self._master.call("NextWork", {
                    "worker_id": self.worker_id,
                    "max_tasks": self._instances}, timeout=30)
            except RpcError:
                time.sleep(min(backoff, 2.0))
                backoff *= 1.5
                continue
            backoff = 0.05
            if resp["job_id"] < 0 or not resp["tasks"]:
                time.sleep(resp.get("wait", 0.2) or 0.2)
                continue
            for t in resp["tasks"]:
                self._task_q.put((resp["job_id"], t))

    def _loader_loop(self):
        while not self._shutdown.is_set():
            try:
                job_id, t = self._task_q.get(timeout=0.2)
            except queue.Empty:
                continue
            stream, task, start, end = t
            pt, err = None, None
            try:
                self._ensure_job(job_id)
                pt = self._executor.prepare_task(stream, task, start, end)
            except Exception as e:
                err = f"{type(e).__name__}: {e}"
            while not self._shutdown.is_set():
                try:
                    self._prep_q.put((job_id, t, pt, err), timeout=0.2)
                    break
                except queue.Full:
                    continue

    def _instance_loop(self, idx):
        while not self._shutdown.is_set():
            try:
                job_id, t, pt, err = self._prep_q.get(timeout=0.2)
            except queue.Empty:
                continue
            stream, task, start, end = t
            ok = err is None
            try:
                if ok:
                    self._ensure_job(job_id)
                    if pt is not None and self._job_id == job_id:
                        self._executor.process_prepared(idx, pt)
                    else:
                        # job changed between prepare and process (rare):
                        # the prepared buffers are dropped and the task
                        # re-reads its inputs
                        self._executor.process_task(idx, stream, task,
                                                    start, end)
            except Exception as e:
                ok, err = False, f"{type(e).__name__}: {e}"
            try:
                self._master.call("FinishedWork", {
                    "worker_id": self.worker_id, "job_id": job_id,
                    "stream": stream, "task": task, "success": ok,
                    "error": err}, timeout=30)
            except RpcError:
                pass

    def _watchdog(self):
        # self-terminate if the master goes silent (reference:
        # worker.cpp:647-673)
        while not self._shutdown.wait(5.0):
            if time.time() - self._last_poke > WATCHDOG_TIMEOUT:
                os._exit(3)

    def shutdown(self):
        self._shutdown.set()
        self._master.try_call("UnregisterWorker",
                              {"worker_id": self.worker_id}, timeout=5)
        self._server.stop()

    def wait(self):
        self._server.wait()


def start_worker(master_addr, db_path, addr="127.0.0.1:0", block=False,
                 **kw):
    w = WorkerServer(master_addr, db_path, addr, **kw)
    if block:
        w.wait()
    return w


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--master", required=True)
    ap.add_argument("--db-path", required=True)
    ap.add_argument("--addr", default="127.0.0.1:0")
    ap.add_argument("--instances", type=int, default=1)
    ap.add_argument("--gpu-ids", default=None,
                    help="comma-separated; default = all visible GPUs "
                         "(reference: default_machine_params)")
    ap.add_argument("--no-watchdog", action="store_true")
    ap.add_argument("--storage-type", default="posix",
                    help="posix | s3 (object store; see config.py)")
    ap.add_argument("--bucket", default="")
    args = ap.parse_args()
    if args.gpu_ids is None:
        gpu_ids = list(range(_core.gpu_device_count()))
    else:
        gpu_ids = [int(x) for x in args.gpu_ids.split(",") if x != ""]
    start_worker(args.master, args.db_path, args.addr, block=True,
                 pipeline_instances=args.instances, gpu_ids=gpu_ids,
                 watchdog=not args.no_watchdog,
                 storage_type=args.storage_type, bucket=args.bucket)


if __name__ == "__main__":
    main()
