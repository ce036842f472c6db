"""Tutorial 09: storage backends.

The database talks to storage through the StorageBackend interface
(csrc/storage.h — parity: the reference's storehouse with posix/GCS/S3
configs). Besides POSIX, an object-store backend with S3 semantics (flat
keyspace, whole-object PUT, range GET, prefix listing) ships built in,
selectable per client or via ~/.scanner_amd.toml:

    [storage]
    type = "s3"
    bucket = "/mnt/bucket"
    db_path = "db"
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

import scanner_amd as sp

tmp = tempfile.mkdtemp(prefix="scanner_tut09_")
bucket = os.path.join(tmp, "bucket")

# same pipeline, object-store backend
sc = sp.Client(db_path="db", storage_type="s3", bucket=bucket)
frames = np.random.RandomState(0).randint(
    0, 255, size=(8, 64, 64, 3)).astype(np.uint8)
video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="svc")
frame = sc.io.Input([video])
hist = sc.ops.Histogram(frame=frame)
out = sp.NamedStream(sc, "hist")
sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
       cache_mode=sp.CacheMode.Overwrite)
rows = list(out.load())
assert len(rows) == 8

# what the "bucket" looks like: flat url-encoded keys, no directories
keys = sorted(os.listdir(bucket))
print(f"bucket holds {len(keys)} objects, e.g.:")
for k in keys[:4]:
    print("  ", k)
assert all(os.path.isfile(os.path.join(bucket, k)) for k in keys)
print("object-store backend OK")
