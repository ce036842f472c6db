"""Typed column payloads + geometry utilities (parity:
python/scannerpy/types.py type registry and scanner/util/bbox.cpp
best/average NMS, serialize.h BoundingBox ser/de).

A type here is a (pack, unpack) pair keyed by name; ops that emit typed
blobs document their type name and readers use `loads`/`NamedStream.load`
with the matching unpack.
"""
import struct

import numpy as np

_REGISTRY = {}


def register_type(name, pack, unpack):
    _REGISTRY[name] = (pack, unpack)


def dumps(name, value):
    return _REGISTRY[name][0](value)


def loads(name, blob):
    return _REGISTRY[name][1](blob)


class BoundingBox:
    """Axis-aligned box with score and label (serialize.h BoundingBox)."""

    __slots__ = ("x1", "y1", "x2", "y2", "score", "label")
    _FMT = "<5fi"

    def __init__(self, x1, y1, x2, y2, score=1.0, label=0):
        self.x1, self.y1, self.x2, self.y2 = (float(x1), float(y1),
                                              float(x2), float(y2))
        self.score = float(score)
        self.label = int(label)

    def to_bytes(self):
        return struct.pack(self._FMT, self.x1, self.y1, self.x2, self.y2,
                           self.score, self.label)

    @classmethod
    def from_bytes(cls, b):
        return cls(*struct.unpack(cls._FMT, b))

    def __repr__(self):
        return (f"BoundingBox({self.x1:.1f},{self.y1:.1f},"
                f"{self.x2:.1f},{self.y2:.1f},s={self.score:.3f},"
                f"l={self.label})")


def pack_bboxes(boxes):
    return struct.pack("<I", len(boxes)) + b"".join(
        b.to_bytes() for b in boxes)


def unpack_bboxes(blob):
    (n,) = struct.unpack_from("<I", blob, 0)
    sz = struct.calcsize(BoundingBox._FMT)
    return [BoundingBox.from_bytes(blob[4 + i * sz:4 + (i + 1) * sz])
            for i in range(n)]


register_type("BoundingBoxList", pack_bboxes, unpack_bboxes)
register_type(
    "Histogram",
    lambda h: np.asarray(h, np.uint32).tobytes(),
    lambda b: np.frombuffer(b, np.uint32).reshape(3, 256))
register_type(
    "FlowStats",
    lambda s: np.asarray(s, np.float32).tobytes(),
    lambda b: np.frombuffer(b, np.float32))


def _iou_matrix(a, b):
    """IoU between two [N,4] / [M,4] arrays of x1,y1,x2,y2."""
    ax1, ay1, ax2, ay2 = a[:, 0, None], a[:, 1, None], a[:, 2, None], \
        a[:, 3, None]
    bx1, by1, bx2, by2 = b[None, :, 0], b[None, :, 1], b[None, :, 2], \
        b[None, :, 3]
    ix = np.maximum(0.0, np.minimum(ax2, bx2) - np.maximum(ax1, bx1))
    iy = np.maximum(0.0, np.minimum(ay2, by2) - np.maximum(ay1, by1))
    inter = ix * iy
    area_a = (ax2 - ax1) * (ay2 - ay1)
    area_b = (bx2 - bx1) * (by2 - by1)
    return inter / np.maximum(area_a + area_b - inter, 1e-9)


def nms_best(boxes, iou_threshold=0.5):
    """Greedy best-first NMS (parity: bbox.cpp best_nms): keep the highest-
    scoring box, drop overlaps, repeat."""
    if not boxes:
        return []
    arr = np.array([[b.x1, b.y1, b.x2, b.y2] for b in boxes], np.float32)
    scores = np.array([b.score for b in boxes], np.float32)
    order = np.argsort(-scores)
    keep = []
    alive = np.ones(len(boxes), bool)
    for i in order:
        if not alive[i]:
            continue
        keep.append(boxes[i])
        ious = _iou_matrix(arr[i:i + 1], arr)[0]
        alive &= ious < iou_threshold
        alive[i] = False
    return keep


def nms_average(boxes, iou_threshold=0.5):
    """Cluster-average NMS (parity: bbox.cpp average_nms): overlapping
    boxes are merged into their score-weighted average."""
    if not boxes:
        return []
    arr = np.array([[b.x1, b.y1, b.x2, b.y2] for b in boxes], np.float32)
    scores = np.array([b.score for b in boxes], np.float32)
    order = np.argsort(-scores)
    out = []
    alive = np.ones(len(boxes), bool)
    for i in order:
        if not alive[i]:
            continue
        ious = _iou_matrix(arr[i:i + 1], arr)[0]
        cluster = alive & (ious >= iou_threshold)
        cluster[i] = True
        w = scores[cluster]
        coords = (arr[cluster] * w[:, None]).sum(0) / w.sum()
        out.append(BoundingBox(*coords, score=float(scores[cluster].max()),
                               label=boxes[i].label))
        alive &= ~cluster
    return out
