"""ThreadSanitizer gate for the engine's shared concurrency primitives
(BoundedQueue between pipeline stages, Profiler): compiles
tests/cpp/tsan_concurrency.cpp with -fsanitize=thread (plain g++ — the
primitives are HIP-free C++) and requires zero TSAN reports. The reference
ships no race detection at all."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_tsan_queue_profiler(tmp_path):
    src = os.path.join(REPO, "tests", "cpp", "tsan_concurrency.cpp")
    binp = str(tmp_path / "tsan_test")
    r = subprocess.run(
        ["g++", "-O1", "-g", "-std=c++17", "-fsanitize=thread", "-pthread",
         src, "-o", binp],
        capture_output=True, text=True, timeout=300)
    if r.returncode != 0:
        pytest.fail(f"tsan build failed:\n{r.stderr[-2000:]}")
    env = dict(os.environ)
    env["TSAN_OPTIONS"] = "halt_on_error=1 exitcode=66"
    run = subprocess.run([binp], capture_output=True, text=True,
                         env=env, timeout=300)
    out = run.stdout + run.stderr
    assert run.returncode == 0, f"tsan reported races:\n{out[-3000:]}"
    assert "OK" in out
    assert "WARNING: ThreadSanitizer" not in out
