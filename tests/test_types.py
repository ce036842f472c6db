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


def test_nms_best_property():
    """nms_best vs a brute-force O(n^2) greedy model over random box sets
    (hypothesis; reference semantics bbox.cpp best_nms)."""
    from hypothesis import given, settings, strategies as st
    from scanner_amd.types import BoundingBox, nms_best

    def iou(a, b):
        ix = max(0.0, min(a.x2, b.x2) - max(a.x1, b.x1))
        iy = max(0.0, min(a.y2, b.y2) - max(a.y1, b.y1))
        inter = ix * iy
        ua = ((a.x2 - a.x1) * (a.y2 - a.y1) +
              (b.x2 - b.x1) * (b.y2 - b.y1) - inter)
        return inter / ua if ua > 0 else 0.0

    def model(boxes, thr):
        rest = sorted(boxes, key=lambda b: -b.score)
        keep = []
        while rest:
            best = rest.pop(0)
            keep.append(best)
            rest = [b for b in rest if iou(best, b) < thr]
        return keep

    box = st.tuples(st.integers(0, 80), st.integers(0, 80),
                    st.integers(1, 40), st.integers(1, 40),
                    st.integers(0, 1000)).map(
        lambda t: BoundingBox(float(t[0]), float(t[1]),
                              float(t[0] + t[2]), float(t[1] + t[3]),
                              score=t[4] / 1000.0))

    @settings(max_examples=60, deadline=None, derandomize=True)
    @given(boxes=st.lists(box, max_size=25),
           thr=st.sampled_from([0.3, 0.5, 0.7]))
    def run(boxes, thr):
        # distinct scores keep the greedy order deterministic
        seen = set()
        boxes = [b for b in boxes
                 if b.score not in seen and not seen.add(b.score)]
        got = nms_best(boxes, iou_threshold=thr)
        want = model(boxes, thr)
        assert [(b.x1, b.y1, b.x2, b.y2, b.score) for b in got] == \
               [(b.x1, b.y1, b.x2, b.y2, b.score) for b in want]

    run()
