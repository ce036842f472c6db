"""Parser robustness: seeded mutation fuzz of the mp4/Annex-B/SPS layer.

Two tiers (reference parity note: the reference delegates all demux/parse
to ffmpeg and so inherits its hardening — scanner/video/*; our
from-scratch parsers carry their own):

1. Python-level: truncated/mutated valid files through the _core bindings
   must parse or raise — never crash the process (tools/fuzz_parsers.py
   is the long-soak driver; this runs a fixed-seed slice).
2. ASAN/UBSan-level: the same fuzz compiled against the parser TUs with
   -fsanitize=address,undefined (plain g++ — the TUs are HIP-free), which
   also catches SILENT out-of-bounds reads that tier 1 cannot.
"""
import os
import subprocess
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_python_fuzz_slice():
    sys.path.insert(0, os.path.join(REPO, "tools"))
    from fuzz_parsers import run

    stats = run(400, seed=7, verbose=False)
    # sanity that the fuzz actually exercises both outcomes
    for name, (ok, raised) in stats.items():
        assert ok + raised == 400
        assert raised > 0, f"{name}: mutations never rejected?"


def test_asan_parser_fuzz(tmp_path):
    from test_video_ingest import make_annexb, make_mp4, make_sps

    paths = {}
    for name, blob in (
        ("c.mp4", make_mp4(n_frames=8, keyframes=(0, 4))[0]),
        ("c.264", make_annexb(gops=2, frames_per_gop=3)[0]),
        ("c.sps", make_sps(64, 48)),
    ):
        p = tmp_path / name
        p.write_bytes(bytes(blob))
        paths[name] = str(p)

    import hashlib
    import tempfile

    src = os.path.join(REPO, "tests", "cpp", "asan_parsers.cpp")
    csrc = os.path.join(REPO, "scanner_amd", "csrc")
    vid = os.path.join(csrc, "video")
    tus = [os.path.join(vid, f) for f in ("mp4.cpp", "h264.cpp",
                                          "svc_cpu.cpp")]
    tus += [os.path.join(csrc, "metadata.cpp"),
            os.path.join(csrc, "storage.cpp")]
    # the ~20 s sanitizer build dominates suite time — cache the binary
    # keyed by a hash of every input (headers included via the csrc tree)
    h = hashlib.sha256()
    for f in [src] + tus:
        h.update(open(f, "rb").read())
    for root, _, files in os.walk(csrc):
        for f in sorted(files):
            if f.endswith(".h"):
                h.update(open(os.path.join(root, f), "rb").read())
    binp = os.path.join(tempfile.gettempdir(),
                        f"sca_asan_fuzz_{h.hexdigest()[:16]}")
    if not os.path.exists(binp):
        r = subprocess.run(
            ["g++", "-O1", "-g", "-std=c++17",
             "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
             src, *tus, "-o", binp],
            capture_output=True, text=True, timeout=300)
        if r.returncode != 0:
            pytest.fail(f"asan build failed:\n{r.stderr[-2000:]}")

    r = subprocess.run(
        [binp, paths["c.mp4"], paths["c.264"], paths["c.sps"], "1500"],
        capture_output=True, text=True, timeout=600)
    out = r.stdout + r.stderr
    assert r.returncode == 0, f"fuzz binary rc={r.returncode}:\n{out[-3000:]}"
    assert "AddressSanitizer" not in out, out[-3000:]
    assert "runtime error" not in out, out[-3000:]
    assert "asan parser fuzz: OK" in out
