"""Multi-process tests for the RCCL/xGMI frame-shard module
(scanner_amd/parallel.py) — run here over gloo with world_size=2 (the
driver's GPU tier exercises the same code over RCCL via bench.py)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

def _free_port():
    """A fresh OS-assigned port per test: the previous fixed ports hit
    TIME_WAIT when the suite re-runs back-to-back (observed flake)."""
    import socket
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        return str(sk.getsockname()[1])


WORKER = r"""
import os, sys
sys.path.insert(0, os.environ["SCA_REPO"])
import torch
from scanner_amd import parallel

rank, world, device = parallel.init_from_env(device_type="cpu")
assert world == 2

# shard_rows covers all rows exactly once
n = 23
spans = [parallel.shard_rows(n, world, r) for r in range(world)]
rows = [i for s, e in spans for i in range(s, e)]
assert sorted(rows) == list(range(n))
s, e = spans[rank]

# gather_column: per-rank blobs arrive on rank 0 in rank order
blobs = [f"row{i}".encode() for i in range(s, e)]
got = parallel.gather_column(blobs, device, max_bytes=16)  # force rounds
if rank == 0:
    assert got == [f"row{i}".encode() for i in range(n)], got
else:
    assert got is None

# gather_column edge shapes: empty blobs, an empty rank, and a payload
# far larger than max_bytes (multi-chunk wire messages)
edge = ([b"", b"x" * 1000, b"", b"tail"] if rank == 0 else [])
got = parallel.gather_column(edge, device, max_bytes=64)
if rank == 0:
    assert got == [b"", b"x" * 1000, b"", b"tail"], got
edge2 = [bytes([rank]) * (300 + rank)]
got = parallel.gather_column(edge2, device, max_bytes=128)
if rank == 0:
    assert got == [b"\x00" * 300, b"\x01" * 301], [len(g) for g in got]

# broadcast_blob
blob = b"weights-payload" if rank == 0 else None
out = parallel.broadcast_blob(blob, device)
assert out == b"weights-payload"

# allreduce_max_time
t = parallel.allreduce_max_time(1.0 + rank, device)
assert t == 2.0
print(f"rank {rank} OK", flush=True)
"""


def test_parallel_gloo_world2(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    procs = []
    port = _free_port()
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "SCA_REPO": REPO,
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": port,
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=120)
        outs.append(out.decode())
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {rank} failed:\n{out}"
        assert f"rank {rank} OK" in out


def test_bench_cpu_world2(tmp_path):
    """bench.py under the driver's multi-rank contract, CPU/gloo: two
    ranks, tiny step counts; rank 0 must print one JSON line with the
    whole-job aggregate."""
    import json
    procs = []
    outs = []
    port = _free_port()
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": port,
        })
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "bench.py"), "--pipeline",
             "hist", "--steps", "1", "--warmup", "0", "--frames-per-step",
             "8"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        out, err = p.communicate(timeout=300)
        outs.append((p.returncode, out.decode(), err.decode()))
    for rank, (rc, out, err) in enumerate(outs):
        assert rc == 0, f"rank {rank}: {err[-2000:]}"
    result = json.loads(outs[0][1].strip().splitlines()[-1])
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "frame-shard dp2"
    assert result["value"] > 0
    # only rank 0 prints the result line (gloo chatter may appear)
    assert not any(ln.startswith("{") for ln in outs[1][1].splitlines())


WORKER8 = r"""
import os, sys, tempfile
sys.path.insert(0, os.environ["SCA_REPO"])
import numpy as np
import torch
from scanner_amd import parallel

rank, world, device = parallel.init_from_env(device_type="cpu")
assert world == 8

import scanner_amd as sp

# Frame-shard DP as an engine mode (SURVEY 2.7): one logical 64-frame clip
# sharded across 8 ranks; each rank runs the engine on its shard and the
# per-row result columns come back to rank 0 over the collective plane
# (gloo here, RCCL/xGMI on GPU boxes).
n_total, h, w = 64, 48, 64
yy, xx = np.mgrid[0:h, 0:w]
all_frames = np.zeros((n_total, h, w, 3), np.uint8)
for i in range(n_total):
    all_frames[i, :, :, 0] = (xx + 2 * i) % 256
    all_frames[i, :, :, 1] = (yy + i) % 256
    all_frames[i, :, :, 2] = (xx + yy + 3 * i) % 256

s, e = parallel.shard_rows(n_total, world, rank)
tmp = tempfile.mkdtemp(prefix=f"sca_dp8_r{rank}_")
sc = sp.Client(db_path=os.path.join(tmp, "db"))
video = sp.NamedVideoStream(sc, "shard", frames=all_frames[s:e],
                            codec="svc")
frame = sc.io.Input([video])
hist = sc.ops.Histogram(frame=frame)
out = sp.NamedStream(sc, "shard_out")
sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
       cache_mode=sp.CacheMode.Overwrite)
blobs = list(out.load())
assert len(blobs) == e - s

got = parallel.gather_column(blobs, device)
if rank == 0:
    assert len(got) == n_total
    for i in range(n_total):
        histo = np.frombuffer(got[i], np.uint32).reshape(3, 256)
        expect = np.stack([
            np.bincount(all_frames[i][:, :, c].ravel(), minlength=256)
            for c in range(3)]).astype(np.uint32)
        np.testing.assert_array_equal(histo, expect)
else:
    assert got is None

# uneven/empty payloads: rank k sends k blobs of k*17 bytes (rank 0 none)
blobs = [bytes([rank]) * (rank * 17) for _ in range(rank)]
got = parallel.gather_column(blobs, device, max_bytes=64)
if rank == 0:
    expect = []
    for r in range(world):
        expect.extend([bytes([r]) * (r * 17)] * r)
    assert got == expect, (len(got), len(expect))
print(f"rank {rank} OK8", flush=True)
"""


def test_parallel_gloo_world8_engine_gather(tmp_path):
    """8-rank frame-shard DP: engine jobs on real column data per rank,
    size-exact P2P gather to rank 0."""
    script = tmp_path / "worker8.py"
    script.write_text(WORKER8)
    procs = []
    port = _free_port()
    for rank in range(8):
        env = dict(os.environ)
        env.update({
            "SCA_REPO": REPO,
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "8",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": port,
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=300)
        outs.append(out.decode())
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {rank} failed:\n{out[-3000:]}"
        assert f"rank {rank} OK8" in out
