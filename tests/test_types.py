"""BoundingBox types + NMS (parity: scanner/util/bbox.cpp best/average
NMS, serialize.h BoundingBox, scannerpy types registry)."""
import numpy as np

from scanner_amd import types


def test_bbox_roundtrip():
    b = types.BoundingBox(1, 2, 30, 40, score=0.9, label=3)
    b2 = types.BoundingBox.from_bytes(b.to_bytes())
    assert (b2.x1, b2.y1, b2.x2, b2.y2) == (1, 2, 30, 40)
    assert abs(b2.score - 0.9) < 1e-6 and b2.label == 3

    blob = types.dumps("BoundingBoxList", [b, b2])
    lst = types.loads("BoundingBoxList", blob)
    assert len(lst) == 2 and lst[0].label == 3


def test_nms_best():
    boxes = [
        types.BoundingBox(0, 0, 10, 10, score=0.9),
        types.BoundingBox(1, 1, 11, 11, score=0.8),   # overlaps first
        types.BoundingBox(50, 50, 60, 60, score=0.7),  # separate
    ]
    kept = types.nms_best(boxes, iou_threshold=0.5)
    assert len(kept) == 2
    assert kept[0].score == 0.9 and kept[1].score == 0.7


def test_nms_average():
    boxes = [
        types.BoundingBox(0, 0, 10, 10, score=1.0),
        types.BoundingBox(2, 2, 12, 12, score=1.0),
        types.BoundingBox(50, 50, 60, 60, score=0.5),
    ]
    merged = types.nms_average(boxes, iou_threshold=0.3)
    assert len(merged) == 2
    m = merged[0]
    assert abs(m.x1 - 1.0) < 1e-5 and abs(m.x2 - 11.0) < 1e-5


def test_histogram_type():
    h = np.arange(3 * 256, dtype=np.uint32).reshape(3, 256)
    blob = types.dumps("Histogram", h)
    np.testing.assert_array_equal(types.loads("Histogram", blob), h)
