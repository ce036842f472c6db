"""Seeded mutation fuzz of the mp4/Annex-B/SPS parsers.

Every iteration mutates or truncates a valid synthesized file and feeds
it to the pure-parsing layer (csrc/video/mp4.cpp, h264.cpp). The parsers
must either succeed or raise a Python exception — a process crash (OOB
read/write, infinite loop) fails the run. tests/test_video_ingest.py runs
a small fixed-seed slice of this as a regression test; this standalone
driver takes --iters for longer soaks.
"""
import argparse
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))

from scanner_amd import _core  # noqa: E402
from test_video_ingest import make_annexb, make_mp4, make_sps  # noqa: E402


def fuzz_one(rng, base, parse):
    data = bytearray(base)
    mode = rng.randrange(3)
    if mode == 0:  # truncate
        data = data[: rng.randrange(len(data) + 1)]
    elif mode == 1:  # mutate 1-8 random bytes
        for _ in range(rng.randrange(1, 9)):
            data[rng.randrange(len(data))] = rng.randrange(256)
    else:  # truncate AND mutate
        data = data[: rng.randrange(1, len(data) + 1)]
        if data:
            for _ in range(rng.randrange(1, 5)):
                data[rng.randrange(len(data))] = rng.randrange(256)
    try:
        parse(bytes(data))
        return "ok"
    except Exception:
        return "raised"


def run(iters, seed=0, verbose=True):
    rng = random.Random(seed)
    mp4 = make_mp4(n_frames=8, keyframes=(0, 4))[0]
    annexb = make_annexb(gops=2, frames_per_gop=3)[0]
    sps = make_sps(64, 48)
    corpora = [
        ("mp4", mp4, _core.mp4_probe),
        ("annexb", annexb, _core.h264_index),
        ("sps", sps, _core.h264_parse_sps_py),
    ]
    stats = {}
    for name, base, parse in corpora:
        ok = raised = 0
        for _ in range(iters):
            if fuzz_one(rng, base, parse) == "ok":
                ok += 1
            else:
                raised += 1
        stats[name] = (ok, raised)
        if verbose:
            print(f"{name}: {iters} iters, {ok} parsed, {raised} raised")
    return stats


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=2000)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    run(args.iters, args.seed)
    print("fuzz: no crashes")
