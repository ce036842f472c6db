"""Bare-metal cluster bootstrap over ssh (parity: the reference client's
start_master/start_workers remote launch, python/scannerpy/client.py:
596-783, which runs `python -c 'import scannerpy; ...start_master()'` on
each host over ssh).

`bootstrap_cluster` starts a master on one host and a worker on each
worker host, over a shared filesystem (db_path must be reachable from
every host — same contract as the reference's shared storage). The ssh
transport is injectable (`ssh_cmd`) so tests exercise the full launch
path against localhost without sshd."""
import shlex
import subprocess
import sys
import time

from .common import ScannerException
from .rpc import RpcClient


class RemoteProcess:
    """One remotely launched process: host + pid, stoppable over the same
    transport."""

    def __init__(self, host, pid, ssh_cmd):
        self.host = host
        self.pid = pid
        self._ssh_cmd = ssh_cmd

    def stop(self, sig="TERM"):
        _run_remote(self._ssh_cmd, self.host,
                    f"kill -{sig} {self.pid} 2>/dev/null || true")


def _run_remote(ssh_cmd, host, command):
    """Run `command` on `host`; returns stdout. ssh_cmd=None uses ssh with
    BatchMode (no password prompts); tests pass ["bash", "-lc"] plus
    host="" to execute locally."""
    if ssh_cmd is None:
        argv = ["ssh", "-o", "BatchMode=yes",
                "-o", "StrictHostKeyChecking=accept-new", host, command]
    else:
        argv = list(ssh_cmd) + ([host, command] if host else [command])
    p = subprocess.run(argv, capture_output=True, text=True, timeout=60)
    if p.returncode != 0:
        raise ScannerException(
            f"remote launch on '{host or 'localhost'}' failed "
            f"(rc={p.returncode}): {p.stderr.strip()[-500:]}")
    return p.stdout


def _launch(ssh_cmd, host, module, args, log_path, python):
    inner = (f"nohup {shlex.quote(python)} -m {module} "
             + " ".join(shlex.quote(a) for a in args)
             + f" > {shlex.quote(log_path)} 2>&1 & echo $!")
    out = _run_remote(ssh_cmd, host, inner).strip().splitlines()
    if not out or not out[-1].isdigit():
        raise ScannerException(
            f"could not read remote pid from '{host}': {out}")
    return RemoteProcess(host, int(out[-1]), ssh_cmd)


class Cluster:
    def __init__(self, master_proc, worker_procs, master_addr):
        self.master = master_proc
        self.workers = worker_procs
        self.master_addr = master_addr

    def stop(self):
        for w in self.workers:
            w.stop()
        # master last: workers unregister while it is still up
        self.master.stop()


def bootstrap_cluster(db_path, master_host, worker_hosts, master_port=5001,
                      instances_per_worker=1, python=None, ssh_cmd=None,
                      log_dir="/tmp", connect_timeout=30.0,
                      master_advertise=None, storage_type="posix",
                      bucket=""):
    """Start master + workers over ssh; returns a Cluster handle whose
    master_addr plugs straight into Client(master=...).

    db_path must be a path valid on every host (shared filesystem), the
    same contract as the reference's shared storage. storage_type="s3"
    (with bucket) runs the whole cluster on the object-store backend
    instead — then db_path is a key prefix, not a filesystem path."""
    python = python or sys.executable
    master_addr = f"{master_advertise or master_host}:{master_port}"
    master = _launch(
        ssh_cmd, master_host, "scanner_amd.master",
        ["--db-path", db_path, "--addr", f"0.0.0.0:{master_port}",
         "--storage-type", storage_type, "--bucket", bucket],
        f"{log_dir}/scanner_master.log", python)

    # wait for the master to answer before launching workers
    rpc = RpcClient(master_addr)
    t0 = time.time()
    while True:
        if rpc.try_call("Ping", {}, timeout=2) is not None:
            break
        if time.time() - t0 > connect_timeout:
            master.stop("KILL")
            raise ScannerException(
                f"master on {master_host} did not answer within "
                f"{connect_timeout}s (see {log_dir}/scanner_master.log)")
        time.sleep(0.2)

    workers = []
    try:
        for i, host in enumerate(worker_hosts):
            workers.append(_launch(
                ssh_cmd, host, "scanner_amd.worker",
                ["--master", master_addr, "--db-path", db_path,
                 "--instances", str(instances_per_worker),
                 "--storage-type", storage_type, "--bucket", bucket],
                f"{log_dir}/scanner_worker_{i}.log", python))
        # wait until every worker registered
        t0 = time.time()
        while True:
            st = rpc.try_call("JobStatus", {"job_id": -1}, timeout=5)
            n = st.get("n_workers", 0) if st else 0
            if n >= len(worker_hosts):
                break
            if time.time() - t0 > connect_timeout:
                raise ScannerException(
                    f"only {n}/{len(worker_hosts)} workers registered "
                    f"within {connect_timeout}s")
            time.sleep(0.2)
    except Exception:
        for w in workers:
            w.stop("KILL")
        master.stop("KILL")
        raise
    return Cluster(master, workers, master_addr)
