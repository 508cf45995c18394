"""SequenceFile image source (reference: SeqImageDataSource.scala)."""

from __future__ import annotations

import glob
import os
from typing import Iterator

from ..proto import caffe_pb
from .image_source import ImageDataSource, ImageSample
from .seqfile import SequenceFileReader


class SeqImageDataSource(ImageDataSource):
    def init(self) -> None:
        from ..utils.fsio import ensure_local
        path = ensure_local(self.source_path)
        if os.path.isdir(path):
            self.files = sorted(
                f for f in glob.glob(os.path.join(path, "*"))
                if not os.path.basename(f).startswith((".", "_")))
        else:
            self.files = [path]

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[ImageSample]:
        return self.persisted_epochs(lambda: self._epoch(rank, world),
                                     epochs)

    def _epoch(self, rank: int, world: int) -> Iterator[ImageSample]:
        from .javaser import key_id_label
        i = 0
        for f in self.files:
            for key, raw in SequenceFileReader(f).items():
                if i % world == rank:
                    jk = key_id_label(key)
                    if jk is not None:
                        # reference-written file (Binary2Sequence.scala):
                        # key = java-serialized (id, label) tuple, value
                        # = raw encoded image bytes
                        sid, label = jk
                        yield ImageSample(sid, label, 3, 0, 0, True,
                                          bytes(raw))
                    else:
                        d = caffe_pb.Datum.FromString(raw)
                        yield ImageSample(key.decode(), d.label, d.channels,
                                          d.height, d.width,
                                          bool(d.encoded), bytes(d.data))
                i += 1
