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
        from ..utils.fsio import ensure_local
        path = ensure_local(self.source_path)
        self.reader = LmdbReader(path)

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[ImageSample]:
        return self.persisted_epochs(lambda: self._epoch(rank, world),
                                     epochs)

    def _epoch(self, rank: int, world: int) -> Iterator[ImageSample]:
        """Key-range partitioned read (reference LmdbRDD.scala:41-95):
        -lmdb_partitions (default: world) disjoint ranges, distributed
        round-robin over ranks — each rank's IO touches only its own
        ranges' pages instead of full-scanning and dropping
        (world-1)/world of the rows."""
        n_parts = int(getattr(self.conf, "lmdbPartitions", 0) or 0)
        n_parts = max(n_parts, world, 1)
        ranges = self.reader.partition_ranges(n_parts)
        for ri in range(rank, len(ranges), world):
            start, end = ranges[ri]
            for key, raw in self.reader.items_range(start, end):
                yield self._decode_datum(key, raw)

    def _decode_datum(self, key: bytes, raw: bytes):
        d = caffe_pb.Datum.FromString(raw)
        if d.float_data:
            import numpy as np
            data = np.asarray(list(d.float_data),
                              dtype=np.float32).tobytes()
            return FloatSample(key.decode(), d.label, d.channels,
                               d.height, d.width, data)
        return ImageSample(key.decode(), d.label, d.channels,
                           d.height, d.width, bool(d.encoded),
                           bytes(d.data))


class FloatSample(ImageSample):
    def __init__(self, id, label, channels, height, width, data):
        super().__init__(id, label, channels, height, width, False, data)
        self.is_float = True
