"""Distributed master/worker tests — real processes/threads, shared temp
storage, no mocks (parity: reference fault-tolerance suite
py_test.py:768-1060: worker SIGKILL mid-job, job blacklisting, no-worker
timeout, late worker join)."""
import os
import signal
import subprocess
import sys
import time

import numpy as np
import pytest

import scanner_amd as sp
from scanner_amd.master import MasterServer
from scanner_amd.worker import start_worker
from conftest import make_video

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def ref_histogram(frame):
    return np.stack([np.bincount(frame[:, :, c].ravel(), minlength=256)
                     for c in range(3)]).astype(np.uint32)


def _mk_db(tmp_path):
    db = str(tmp_path / "db")
    os.makedirs(db, exist_ok=True)
    return db


def spawn_worker_proc(master_addr, db_path, instances=1):
    """Real OS process (reference: tests/spawn_worker.py)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.Popen(
        [sys.executable, "-m", "scanner_amd.worker", "--master",
         master_addr, "--db-path", db_path, "--instances", str(instances),
         "--no-watchdog"],
        env=env, start_new_session=True)


def test_distributed_histogram(tmp_path):
    db = _mk_db(tmp_path)
    master = MasterServer(db)
    workers = [start_worker(master.addr, db) for _ in range(2)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        vids = [make_video(n=12, seed=s) for s in range(2)]
        streams = [sp.NamedVideoStream(sc, f"d{i}", frames=v, codec="raw")
                   for i, v in enumerate(vids)]
        frame = sc.io.Input(streams)
        hist = sc.ops.Histogram(frame=frame)
        outs = [sp.NamedStream(sc, f"d{i}_hist") for i in range(2)]
        sc.run(sc.io.Output(hist, outs), sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
        for i, o in enumerate(outs):
            rows = list(o.load())
            assert len(rows) == 12
            for r, blob in enumerate(rows):
                got = np.frombuffer(blob, np.uint32).reshape(3, 256)
                np.testing.assert_array_equal(got, ref_histogram(vids[i][r]))
        sc.shutdown()
    finally:
        for w in workers:
            w.shutdown()
        master.shutdown()


def test_fault_tolerance_worker_kill(tmp_path):
    db = _mk_db(tmp_path)
    master = MasterServer(db, task_timeout=20)
    procs = [spawn_worker_proc(master.addr, db) for _ in range(2)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        n = 40
        tab = sc.new_table("ft", ["col"],
                           [[int(i).to_bytes(8, "little")] for i in range(n)])
        col = sc.io.Input([tab])
        slow = sc.ops.Sleep(ignore=col, ms=60)
        out = sp.NamedStream(sc, "ft_out")
        sink = sc.io.Output(slow, [out])

        import threading
        done = {}

        def run():
            try:
                sc.run(sink, sp.PerfParams.manual(1, 2),
                       cache_mode=sp.CacheMode.Overwrite)
                done["ok"] = True
            except Exception as e:
                done["err"] = e

        th = threading.Thread(target=run)
        th.start()
        time.sleep(1.5)  # mid-job
        os.killpg(procs[0].pid, signal.SIGKILL)  # hard-kill one worker
        th.join(timeout=120)
        assert not th.is_alive(), "job did not complete after worker kill"
        assert done.get("ok"), f"job failed: {done.get('err')}"
        vals = [int.from_bytes(b, "little") for b in out.load()]
        assert vals == list(range(n))  # Sleep copies its input through
        sc.shutdown()
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except Exception:
                pass
        master.shutdown()


def test_job_blacklist(tmp_path):
    db = _mk_db(tmp_path)

    @sp.register_python_op(name="AlwaysFails")
    def AlwaysFails(col: bytes) -> bytes:
        raise RuntimeError("poison")

    master = MasterServer(db, task_timeout=20)
    worker = start_worker(master.addr, db)
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        tab = sc.new_table("bl", ["col"],
                           [[int(i).to_bytes(8, "little")] for i in range(4)])
        col = sc.io.Input([tab])
        bad = sc.ops.AlwaysFails(col=col)
        out = sp.NamedStream(sc, "bl_out")
        with pytest.raises(sp.ScannerException, match="blacklist"):
            sc.run(sc.io.Output(bad, [out]), sp.PerfParams.manual(2, 4),
                   cache_mode=sp.CacheMode.Overwrite)
        sc.shutdown()
    finally:
        worker.shutdown()
        master.shutdown()


def test_no_workers_timeout(tmp_path):
    db = _mk_db(tmp_path)
    master = MasterServer(db, no_workers_timeout=2.0)
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        tab = sc.new_table("nw", ["col"], [[b"x"]])
        col = sc.io.Input([tab])
        inc = sc.ops.TestIncrement(ignore=col)
        out = sp.NamedStream(sc, "nw_out")
        with pytest.raises(sp.ScannerException, match="no workers"):
            sc.run(sc.io.Output(inc, [out]), sp.PerfParams.manual(1, 1),
                   cache_mode=sp.CacheMode.Overwrite)
        sc.shutdown()
    finally:
        master.shutdown()


def test_late_worker_join(tmp_path):
    db = _mk_db(tmp_path)
    master = MasterServer(db, no_workers_timeout=30.0)
    worker_holder = {}
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        n = 8
        tab = sc.new_table("lw", ["col"],
                           [[int(i).to_bytes(8, "little")] for i in range(n)])
        col = sc.io.Input([tab])
        inc = sc.ops.TestIncrement(ignore=col)
        out = sp.NamedStream(sc, "lw_out")
        sink = sc.io.Output(inc, [out])

        import threading

        def join_late():
            time.sleep(1.0)
            worker_holder["w"] = start_worker(master.addr, db)

        threading.Thread(target=join_late, daemon=True).start()
        sc.run(sink, sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
        assert len(list(out.load())) == n
        sc.shutdown()
    finally:
        if "w" in worker_holder:
            worker_holder["w"].shutdown()
        master.shutdown()


def test_distributed_profile_collection(tmp_path):
    """Workers write per-node profiler files; client gets a merged Profile
    (parity: jobs/<id>/profile_<node>.bin + scannerpy Profile)."""
    db = _mk_db(tmp_path)
    master = MasterServer(db)
    workers = [start_worker(master.addr, db) for _ in range(2)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        frames = make_video(n=12)
        video = sp.NamedVideoStream(sc, "dp", frames=frames, codec="raw")
        frame = sc.io.Input([video])
        hist = sc.ops.Histogram(frame=frame)
        out = sp.NamedStream(sc, "dp_hist")
        prof = sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
                      cache_mode=sp.CacheMode.Overwrite)
        assert prof is not None
        stats = prof.statistics()
        assert any(k.startswith("op:Histogram") for k in stats), stats
        trace = prof.write_trace(str(tmp_path / "trace.json"))
        import json
        with open(trace) as f:
            assert len(json.load(f)["traceEvents"]) > 0
    finally:
        for w in workers:
            w.shutdown()
        master.shutdown()


def test_bulk_job_queueing(tmp_path):
    """A second bulk job submitted while one runs is queued and executes
    after it (reference: master job_processor queue)."""
    db = _mk_db(tmp_path)
    master = MasterServer(db)
    workers = [start_worker(master.addr, db)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        vids = [make_video(n=10, seed=s) for s in range(2)]
        streams = [sp.NamedVideoStream(sc, f"q{i}", frames=v, codec="raw")
                   for i, v in enumerate(vids)]
        results = {}

        def run_one(i):
            try:
                sc2 = sp.Client(db_path=db, master=master.addr)
                frame = sc2.io.Input([streams[i]])
                hist = sc2.ops.Histogram(frame=frame)
                out = sp.NamedStream(sc2, f"q{i}_hist")
                sc2.run(sc2.io.Output(hist, [out]),
                        sp.PerfParams.manual(2, 4),
                        cache_mode=sp.CacheMode.Overwrite)
                results[i] = len(list(out.load()))
            except Exception as e:
                results[i] = f"{type(e).__name__}: {e}"

        import threading
        ts = [threading.Thread(target=run_one, args=(i,)) for i in range(2)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=120)
        assert results == {0: 10, 1: 10}, results
    finally:
        for w in workers:
            w.shutdown()
        master.shutdown()


def test_straggler_task_timeout(tmp_path):
    """Tasks exceeding the master's task timeout are failed and retried;
    repeated timeouts blacklist the stream (reference: per-task timeout
    master.cpp:1750-1776 + blacklist)."""
    db = _mk_db(tmp_path)
    master = MasterServer(db, task_timeout=1.0)
    workers = [start_worker(master.addr, db)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        tab = sc.new_table("slow", ["col"],
                           [[int(i).to_bytes(4, "little")] for i in range(4)])
        col = sc.io.Input([tab])
        slow = sc.ops.Sleep(ignore=col, ms=4000)
        out = sp.NamedStream(sc, "slow_out")
        with pytest.raises(sp.ScannerException, match="blacklist"):
            sc.run(sc.io.Output(slow, [out]), sp.PerfParams.manual(2, 4),
                   cache_mode=sp.CacheMode.Overwrite)
    finally:
        for w in workers:
            w.shutdown()
        master.shutdown()


def test_master_stress_8workers_2jobs_churn(tmp_path):
    """8 workers, 2 queued bulk jobs, kill 2 workers mid-flight and join 2
    late ones — both jobs must complete with correct results (VERDICT r01
    #9 master hardening: analysis off the RPC lock, concurrent pings)."""
    import threading
    db = _mk_db(tmp_path)
    # generous timeouts: this test runs 10 OS processes and flaked under
    # heavy machine load with tighter settings
    master = MasterServer(db, task_timeout=60)
    procs = [spawn_worker_proc(master.addr, db) for _ in range(8)]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        n = 48
        vids = [make_video(n=n, seed=s) for s in range(2)]
        streams = [sp.NamedVideoStream(sc, f"st{i}", frames=v, codec="raw")
                   for i, v in enumerate(vids)]

        results = {}

        def run_job(tag):
            # separate client per thread (its own RPC connection)
            try:
                scj = sp.Client(db_path=db, master=master.addr)
                frame = scj.io.Input(streams)
                slow = scj.ops.Sleep(ignore=frame, ms=20)
                outs = [sp.NamedStream(scj, f"o{tag}_{i}") for i in range(2)]
                scj.run(scj.io.Output(slow, outs),
                        sp.PerfParams.manual(2, 4),
                        cache_mode=sp.CacheMode.Overwrite)
                results[tag] = [list(o.load()) for o in outs]
            except Exception as e:  # surfaced by the assertion below
                results[tag] = e

        threads = [threading.Thread(target=run_job, args=(t,))
                   for t in range(2)]
        for t in threads:
            t.start()
        # churn: kill two workers mid-flight, join two new ones
        time.sleep(1.5)
        for p in procs[:2]:
            os.killpg(p.pid, signal.SIGKILL)
        procs.extend(spawn_worker_proc(master.addr, db) for _ in range(2))
        for t in threads:
            t.join(timeout=300)
            assert not t.is_alive(), "job thread hung"
        for tag in range(2):
            assert not isinstance(results.get(tag), Exception), \
                f"job {tag} failed: {results[tag]}"
            assert len(results[tag]) == 2
            for rows in results[tag]:
                assert len(rows) == n
        sc.shutdown()
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
        master.shutdown()


def test_cluster_bootstrap_localhost(tmp_path):
    """ssh-style cluster bootstrap (parity: reference client.py:596-783)
    with the transport pointed at local bash — exercises the full remote
    launch path (nohup, pid capture, readiness polling, teardown) without
    sshd."""
    from scanner_amd.cluster import bootstrap_cluster
    db = _mk_db(tmp_path)
    os.environ.setdefault("PYTHONPATH", "")
    old_pp = os.environ["PYTHONPATH"]
    os.environ["PYTHONPATH"] = REPO + os.pathsep + old_pp
    try:
        import socket
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        cluster = bootstrap_cluster(
            db, master_host="", worker_hosts=["", ""],
            master_port=port, master_advertise="127.0.0.1",
            ssh_cmd=["bash", "-c"], log_dir=str(tmp_path))
        try:
            sc = sp.Client(db_path=db, master=cluster.master_addr)
            vid = make_video(n=10)
            stream = sp.NamedVideoStream(sc, "cb", frames=vid, codec="raw")
            frame = sc.io.Input([stream])
            hist = sc.ops.Histogram(frame=frame)
            out = sp.NamedStream(sc, "cb_out")
            sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
                   cache_mode=sp.CacheMode.Overwrite)
            rows = list(out.load())
            assert len(rows) == 10
            for r, blob in enumerate(rows):
                got = np.frombuffer(blob, np.uint32).reshape(3, 256)
                np.testing.assert_array_equal(got, ref_histogram(vid[r]))
            sc.shutdown()
        finally:
            cluster.stop()
    finally:
        os.environ["PYTHONPATH"] = old_pp


def test_distributed_on_object_store(tmp_path, monkeypatch):
    """Full master + worker-subprocess job over the S3-semantics object
    store: every table/metadata access on master, worker, and client goes
    through the flat-keyspace backend (the kube + cloud-storage deployment
    shape; profile files remain posix debug artifacts). The worker runs as
    a REAL OS process with --storage-type s3."""
    monkeypatch.chdir(tmp_path)  # posix profile artifacts land here
    db = "db"
    bucket = str(tmp_path / "bucket")
    master = MasterServer(db, storage_type="s3", bucket=bucket)
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    wp = subprocess.Popen(
        [sys.executable, "-m", "scanner_amd.worker", "--master",
         master.addr, "--db-path", db, "--no-watchdog",
         "--storage-type", "s3", "--bucket", bucket],
        env=env, start_new_session=True)
    try:
        sc = sp.Client(db_path=db, master=master.addr,
                       storage_type="s3", bucket=bucket)
        vid = make_video(n=10, seed=3)
        video = sp.NamedVideoStream(sc, "os_d", frames=vid, codec="svc")
        frame = sc.io.Input([video])
        hist = sc.ops.Histogram(frame=frame)
        out = sp.NamedStream(sc, "os_d_hist")
        sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
        rows = list(out.load())
        assert len(rows) == 10
        for r, blob in enumerate(rows):
            got = np.frombuffer(blob, np.uint32).reshape(3, 256)
            np.testing.assert_array_equal(got, ref_histogram(vid[r]))
        # table data + metadata live in the flat bucket; the posix db path
        # holds at most jobs/ profile debug artifacts
        assert os.listdir(bucket)
        posix_db = os.path.join(os.getcwd(), db)
        if os.path.exists(posix_db):
            assert set(os.listdir(posix_db)) <= {"jobs"}, \
                os.listdir(posix_db)
        sc.shutdown()
    finally:
        if wp.poll() is None:
            os.killpg(wp.pid, signal.SIGKILL)
        wp.wait()
        master.shutdown()


def test_worker_multiple_pipeline_instances(tmp_path):
    """One worker running 3 pipeline instances (the 8-GPU node shape:
    --instances = gpus_per_node, each instance a full engine replica) —
    tasks fan out across instances and results stay exact."""
    db = _mk_db(tmp_path)
    master = MasterServer(db)
    wp = spawn_worker_proc(master.addr, db, instances=3)
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        vid = make_video(n=24, seed=9)
        video = sp.NamedVideoStream(sc, "mi", frames=vid, codec="svc")
        frame = sc.io.Input([video])
        hist = sc.ops.Histogram(frame=frame)
        out = sp.NamedStream(sc, "mi_hist")
        sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(2, 4),
               cache_mode=sp.CacheMode.Overwrite)
        rows = list(out.load())
        assert len(rows) == 24
        for r, blob in enumerate(rows):
            got = np.frombuffer(blob, np.uint32).reshape(3, 256)
            np.testing.assert_array_equal(got, ref_histogram(vid[r]))
        sc.shutdown()
    finally:
        if wp.poll() is None:
            os.killpg(wp.pid, signal.SIGKILL)
        wp.wait()
        master.shutdown()


@pytest.mark.parametrize("seed", [3, 11])
def test_scheduler_chaos_seeded(tmp_path, seed):
    """Seeded randomized churn: workers killed and joined at random times
    while a Sleep-op job runs; the job must complete with exact results
    whatever the interleaving (generalizes the fixed-timing churn test)."""
    import random
    import threading
    rng = random.Random(seed)
    db = _mk_db(tmp_path)
    master = MasterServer(db, task_timeout=20)
    procs = [spawn_worker_proc(master.addr, db)
             for _ in range(rng.randint(2, 4))]
    try:
        sc = sp.Client(db_path=db, master=master.addr)
        n = 30
        tab = sc.new_table("ch", ["col"],
                           [[int(i).to_bytes(8, "little")]
                            for i in range(n)])
        col = sc.io.Input([tab])
        slow = sc.ops.Sleep(ignore=col, ms=rng.randint(15, 40))
        out = sp.NamedStream(sc, "ch_out")
        done = {}

        def run():
            try:
                sc.run(sc.io.Output(slow, [out]), sp.PerfParams.manual(1, 2),
                       cache_mode=sp.CacheMode.Overwrite)
                done["ok"] = True
            except Exception as e:  # pragma: no cover
                done["err"] = e

        th = threading.Thread(target=run)
        th.start()
        # random churn script: 2-3 events at random offsets
        for _ in range(rng.randint(2, 3)):
            time.sleep(rng.uniform(0.2, 0.8))
            if rng.random() < 0.5 and len(procs) > 1:
                victim = procs.pop(rng.randrange(len(procs)))
                try:
                    os.killpg(victim.pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass
            else:
                procs.append(spawn_worker_proc(master.addr, db))
        th.join(timeout=120)
        assert not th.is_alive(), "job hung under churn"
        assert done.get("ok"), f"job failed: {done.get('err')}"
        vals = [int.from_bytes(b, "little") for b in out.load()]
        assert vals == list(range(n))
        sc.shutdown()
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
        master.shutdown()
