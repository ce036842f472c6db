"""Tutorial 06: the DNN model families — classification (ResNet-50),
pose keypoints, and object detection with NMS — all on the MFMA GEMM
path with random-init weights. Requires a GPU; exits cleanly without one.
(Parity: the reference's scannertools DNN examples.)"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import scanner_amd as sp
from scanner_amd import _core, types


def main():
    if not _core.have_gpu():
        print("no GPU visible; skipping model tutorial")
        return
    sc = sp.Client(db_path=tempfile.mkdtemp(prefix="sca_tut06_"))
    frames = np.random.RandomState(6).randint(
        0, 255, size=(8, 480, 640, 3), dtype=np.uint8)
    video = sp.NamedVideoStream(sc, "clip", frames=frames, codec="svc")

    frame = sc.io.Input([video])
    logits = sc.ops.ResNet50(frame=frame, device=sp.DeviceType.GPU)
    pose = sc.ops.Pose(frame=frame, device=sp.DeviceType.GPU)
    boxes = sc.ops.Detector(frame=frame, device=sp.DeviceType.GPU)
    out = sp.NamedStream(sc, "models_out")
    sc.run(sc.io.Output([logits, pose, boxes], [out]),
           sp.PerfParams.estimate(), cache_mode=sp.CacheMode.Overwrite,
           gpu_ids=[0])

    top1 = [int(np.argmax(np.frombuffer(b, np.float32)))
            for b in sp.NamedStream(sc, "models_out",
                                    column="logits").load()]
    kps = [np.frombuffer(b, np.float32).reshape(19, 3)
           for b in sp.NamedStream(sc, "models_out", column="pose").load()]
    dets = [types.loads("BoundingBoxList", b)
            for b in sp.NamedStream(sc, "models_out", column="boxes").load()]
    print("top-1 classes:", top1)
    print("frame 0 keypoints:", kps[0][:3], "...")
    print("frame 0 detections:", len(dets[0]), "boxes")


if __name__ == "__main__":
    main()
