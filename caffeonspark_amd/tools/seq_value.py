"""Helpers shared by the converters."""

from __future__ import annotations

from ..proto import caffe_pb


def datum_from_image_file(path: str, label: float) -> caffe_pb.Datum:
    with open(path, "rb") as fh:
        raw = fh.read()
    d = caffe_pb.Datum()
    d.data = raw
    d.label = int(label)
    d.encoded = True
    return d


def datum_from_array(arr, label: float) -> caffe_pb.Datum:
    """CHW uint8 numpy array -> raw Datum."""
    d = caffe_pb.Datum()
    d.channels, d.height, d.width = arr.shape
    d.data = arr.tobytes()
    d.label = int(label)
    d.encoded = False
    return d
