"""Diagnostic: GPU SVC decode-exact across sizes; report mismatch shape."""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import scanner_amd as sp


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


sc = sp.Client(db_path=os.path.join(tempfile.mkdtemp(), "db"))
for (h, w) in [(120, 160), (360, 640), (1080, 1920), (2160, 3840)]:
    frames = smooth(6, h, w)
    name = f"d{h}"
    video = sp.NamedVideoStream(sc, name, frames=frames, codec="svc")
    fr = sc.io.Input([video])
    out = sp.NamedStream(sc, name + "_o")
    sc.run(sc.io.Output(fr, [out]), sp.PerfParams.manual(2, 4),
           cache_mode=sp.CacheMode.Overwrite, gpu_ids=[0])
    got = np.stack(list(sp.NamedVideoStream(sc, name + "_o").load()))
    flat_g = got.reshape(6, -1)
    flat_r = frames.reshape(6, -1)
    nbytes = flat_r.shape[1]
    ngroups = (nbytes + 31) // 32
    print(f"== {h}x{w} nbytes={nbytes} ngroups={ngroups} "
          f"nsuper={(ngroups+127)//128}")
    for i in range(6):
        mism = np.nonzero(flat_g[i] != flat_r[i])[0]
        if len(mism) == 0:
            print(f"  frame {i}: OK")
        else:
            first = mism[0]
            print(f"  frame {i}: {len(mism)} wrong; first@{first} "
                  f"(group {first//32}, super {first//32//128}, "
                  f"lane {(first//32)%128}, byte {first%32}) "
                  f"got={flat_g[i][first]} want={flat_r[i][first]}")
            # distribution over supergroups
            sgs = np.unique(mism // (32 * 128))
            print(f"    supers affected: {len(sgs)} "
                  f"(first 10: {sgs[:10].tolist()})")
