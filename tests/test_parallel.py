"""Multi-process tests for the RCCL/xGMI frame-shard module
(scanner_amd/parallel.py) — run here over gloo with world_size=2 (the
driver's GPU tier exercises the same code over RCCL via bench.py)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

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
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "SCA_REPO": REPO,
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29611",
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
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29617",
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
