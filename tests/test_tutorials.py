"""Run every examples/ tutorial in a subprocess (parity: reference CI runs
pytest -k test_tutorial over examples/tutorials, py_test.py:47-58)."""
import glob
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TUTORIALS = sorted(glob.glob(os.path.join(REPO, "examples", "*.py")))


@pytest.mark.parametrize("script", TUTORIALS,
                         ids=[os.path.basename(t) for t in TUTORIALS])
def test_tutorial(script):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run([sys.executable, script], env=env,
                       capture_output=True, timeout=300)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
