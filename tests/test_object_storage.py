"""Object-store storage backend (S3 semantics over a local bucket —
csrc/storage.cpp ObjectStorage; parity: the reference's GCS/S3 storehouse
configs, scannerpy config.py:75-89)."""
import os

import numpy as np
import pytest

import scanner_amd as sp
from conftest import make_video


def _mk_client(tmp_path):
    return sp.Client(db_path="db", storage_type="s3",
                     bucket=str(tmp_path / "bucket"))


def test_pipeline_on_object_store(tmp_path):
    sc = _mk_client(tmp_path)
    frames = make_video(n=12, h=32, w=48)
    video = sp.NamedVideoStream(sc, "os_in", frames=frames, codec="svc")
    frame = sc.io.Input([video])
    hist = sc.ops.Histogram(frame=frame)
    out = sp.NamedStream(sc, "os_out")
    sc.run(sc.io.Output(hist, [out]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    rows = list(out.load())
    assert len(rows) == 12
    for i, blob in enumerate(rows):
        got = np.frombuffer(blob, np.uint32).reshape(3, 256)
        expect = np.stack([np.bincount(frames[i][:, :, c].ravel(),
                                       minlength=256)
                           for c in range(3)]).astype(np.uint32)
        np.testing.assert_array_equal(got, expect)

    # the bucket is flat: url-encoded keys, no subdirectories
    bucket = tmp_path / "bucket"
    entries = os.listdir(bucket)
    assert entries, "bucket is empty"
    assert all(os.path.isfile(bucket / e) for e in entries)
    assert any("%2F" in e for e in entries)


def test_object_store_overwrite_and_delete(tmp_path):
    sc = _mk_client(tmp_path)
    frames = make_video(n=4, h=16, w=16)
    sp.NamedVideoStream(sc, "t1", frames=frames, codec="raw")
    assert sc.has_table("t1")
    n_before = len(os.listdir(tmp_path / "bucket"))
    sc.delete_table("t1")
    assert not sc.has_table("t1")
    assert len(os.listdir(tmp_path / "bucket")) < n_before


def test_object_store_reopen(tmp_path):
    sc = _mk_client(tmp_path)
    frames = make_video(n=6, h=16, w=16)
    sp.NamedVideoStream(sc, "persist", frames=frames, codec="svc")
    del sc
    sc2 = _mk_client(tmp_path)
    assert sc2.has_table("persist")
    got = np.stack(list(sp.NamedVideoStream(sc2, "persist").load()))
    np.testing.assert_array_equal(got, frames)


def test_config_toml_selects_object_store(tmp_path):
    cfg = tmp_path / "scanner.toml"
    cfg.write_text(
        '[storage]\ntype = "s3"\n'
        f'bucket = "{tmp_path / "cfg_bucket"}"\n'
        f'db_path = "db"\n')
    sc = sp.Client(config_path=str(cfg))
    frames = make_video(n=3, h=16, w=16)
    sp.NamedVideoStream(sc, "cfg_t", frames=frames, codec="raw")
    assert (tmp_path / "cfg_bucket").is_dir()
    assert len(os.listdir(tmp_path / "cfg_bucket")) > 0


def test_config_rejects_unknown_type(tmp_path):
    cfg = tmp_path / "bad.toml"
    cfg.write_text('[storage]\ntype = "tape"\n')
    with pytest.raises(Exception, match="unknown storage type"):
        sp.Client(config_path=str(cfg))
