"""Tutorial 08: user sources and sinks.

Jobs don't have to start or end at database tables: a registered Source
feeds the graph rows from anywhere (here: one row per file in a
directory), and a registered Sink writes results anywhere (here: one file
per output row). Both are C++ classes behind SCA_REGISTER_SOURCE/SINK
(csrc/ops/source.h — parity: scanner/api/source.h, sink.h,
enumerator.h); `Files` ships built in.
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import scanner_amd as sp

tmp = tempfile.mkdtemp(prefix="scanner_tut08_")
sc = sp.Client(db_path=os.path.join(tmp, "db"))

# a directory of blobs to process
in_dir = os.path.join(tmp, "in")
os.makedirs(in_dir)
paths = []
for i in range(6):
    p = os.path.join(in_dir, f"doc{i}.txt")
    with open(p, "w") as f:
        f.write(f"document {i} " * (i + 1))
    paths.append(p)

# Files source -> op -> Files sink: no table anywhere
col = sc.sources.Files(paths)
doubled = sc.ops.ConcatBytes(inputs=[col, col])
out_dir = os.path.join(tmp, "out")
os.makedirs(out_dir)
sink = sc.sinks.Files(doubled, out_dir, ext="txt")
sc.run(sink, sp.PerfParams.manual(2, 4),
       cache_mode=sp.CacheMode.Overwrite)

written = sorted(os.listdir(out_dir))
print(f"wrote {len(written)} files:", written)
assert len(written) == 6
body = open(os.path.join(out_dir, "c0_2.txt")).read()
assert body == "document 2 " * 3 * 2
print("sources/sinks OK")
