import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from scanner_amd import _core

yy, xx = np.mgrid[0:1080, 0:1920]
rng = np.random.RandomState(0)
tex = rng.randint(0, 32, size=(1080, 1920, 3)).astype(np.int32)
f = np.zeros((1, 1080, 1920, 3), np.uint8)
f[0, :, :, 0] = (xx + tex[:, :, 0]) % 256
f[0, :, :, 1] = (yy + tex[:, :, 1]) % 256
f[0, :, :, 2] = (xx + yy + tex[:, :, 2]) % 256
_core.init_memory(1 << 30, 4 << 30, [0])
r = _core.svc_gpu_debug(f)
dev = r["dev"]
for i, h in enumerate(r["host"]):
    d = dev[i * 8:(i + 1) * 8]
    print(f"s={255+i} host so={h['super_off']} w0={h['w_lane0']} "
          f"q0={h['q0']:#010x} q1={h['q1']:#010x} poff={h['packed_off']}")
    print(f"      dev  so={d[0]} myoff={d[1]} w={d[2]} q0={d[3]:#010x} "
          f"q1={d[4]:#010x} poff={d[5]} pkt0={d[6]} tag={d[7]:#x}")
