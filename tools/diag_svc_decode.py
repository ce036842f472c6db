#!/usr/bin/env python3
"""SVC GPU-decode diagnostics (the tool that isolated the round-2
GOP-batch corruption past byte 2^20 — profiles/r02_results.md update 5).

Modes (run on a GPU box):
  --mode engine     decode through the full engine pipeline at several
                    sizes; reports first mismatching byte/group/supergroup
  --mode direct     encode(CPU) -> svc_gpu_roundtrip (kernel only, no
                    engine/cache/chunking), incl. sparse wants and
                    all-key GOPs
  --mode dump       kernel-side read dump (super_off/widths/packed words)
                    vs host-computed expectations at supergroups 255-257
"""
import argparse
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def smooth(n, h, w):
    yy, xx = np.mgrid[0:h, 0:w]
    rng = np.random.RandomState(0)
    tex = rng.randint(0, 32, size=(h, w, 3)).astype(np.int32)
    f = np.zeros((n, h, w, 3), np.uint8)
    for i in range(n):
        f[i, :, :, 0] = (xx + 2 * i + tex[:, :, 0]) % 256
        f[i, :, :, 1] = (yy + i + tex[:, :, 1]) % 256
        f[i, :, :, 2] = (xx + yy + 3 * i + tex[:, :, 2]) % 256
    return f


def report(tag, got, ref):
    fg, fr = got.reshape(got.shape[0], -1), ref.reshape(ref.shape[0], -1)
    for i in range(fg.shape[0]):
        m = np.nonzero(fg[i] != fr[i])[0]
        if len(m) == 0:
            print(f"  {tag} frame {i}: OK")
        else:
            first = m[0]
            print(f"  {tag} frame {i}: {len(m)} wrong; first@{first} "
                  f"(group {first//32}, super {first//32//128}) "
                  f"got={fg[i][first]} want={fr[i][first]}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="direct",
                    choices=["engine", "direct", "dump"])
    args = ap.parse_args()
    from scanner_amd import _core

    if args.mode == "engine":
        import scanner_amd as sp
        sc = sp.Client(db_path=os.path.join(tempfile.mkdtemp(), "db"))
        for (h, w) in [(120, 160), (360, 640), (1080, 1920), (2160, 3840)]:
            frames = smooth(6, h, w)
            name = f"d{h}"
            video = sp.NamedVideoStream(sc, name, frames=frames,
                                        codec="svc")
            fr = sc.io.Input([video])
            out = sp.NamedStream(sc, name + "_o")
            sc.run(sc.io.Output(fr, [out]), sp.PerfParams.manual(2, 4),
                   cache_mode=sp.CacheMode.Overwrite, gpu_ids=[0])
            got = np.stack(list(sp.NamedVideoStream(sc, name + "_o")
                                .load()))
            print(f"== engine {h}x{w}")
            report("", got, frames)
        return

    _core.init_memory(1 << 30, 4 << 30, [0])
    if args.mode == "direct":
        for (h, w) in [(360, 640), (1080, 1920), (2160, 3840)]:
            print(f"== direct {h}x{w}")
            frames = smooth(4, h, w)
            report("gop16", _core.svc_gpu_roundtrip(frames, 16, []), frames)
            report("allkey", _core.svc_gpu_roundtrip(frames, 1, []), frames)
            got = _core.svc_gpu_roundtrip(frames, 16, [3])
            report("want[3]", got, frames[3:4])
        return

    # dump
    frames = smooth(1, 1080, 1920)
    r = _core.svc_gpu_debug(frames)
    dev = r["dev"]
    for i, hh in enumerate(r["host"]):
        d = dev[i * 8:(i + 1) * 8]
        print(f"s={255+i} host so={hh['super_off']} w0={hh['w_lane0']} "
              f"q0={hh['q0']:#010x} q1={hh['q1']:#010x}")
        print(f"      dev  so={d[0]} myoff={d[1]} w={d[2]} q0={d[3]:#010x} "
              f"q1={d[4]:#010x} tag={d[7]:#x}")


if __name__ == "__main__":
    main()
