"""LMDB-backed image source (reference: LMDB.scala + LmdbRDD.scala).

Reads Caffe-standard LMDB databases of `Datum` binary protos (the format
produced by caffe's convert-dataset tools and by our tools/).  Partitioning
is round-robin over entries by (rank, world) — the analog of LmdbRDD's
key-range partitions.
"""

from __future__ import annotations

from typing import Iterator

from ..proto import caffe_pb
from .image_source import ImageDataSource, ImageSample
from .lmdb_io import LmdbReader


class LMDBSource(ImageDataSource):
    def init(self) -> None:
        path = self.source_path
        if path.startswith("file:"):
            path = path[5:]
        self.reader = LmdbReader(path)

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[ImageSample]:
        return self.persisted_epochs(lambda: self._epoch(rank, world),
                                     epochs)

    def _epoch(self, rank: int, world: int) -> Iterator[ImageSample]:
        for i, (key, raw) in enumerate(self.reader.items()):
            if i % world != rank:
                continue
            d = caffe_pb.Datum.FromString(raw)
            if d.float_data:
                import numpy as np
                data = np.asarray(list(d.float_data),
                                  dtype=np.float32).tobytes()
                yield FloatSample(key.decode(), d.label, d.channels,
                                  d.height, d.width, data)
            else:
                yield ImageSample(key.decode(), d.label, d.channels,
                                  d.height, d.width, bool(d.encoded),
                                  bytes(d.data))


class FloatSample(ImageSample):
    def __init__(self, id, label, channels, height, width, data):
        super().__init__(id, label, channels, height, width, False, data)
        self.is_float = True
