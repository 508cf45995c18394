"""DataFrame (parquet) sources.

ImageDataFrameSource — reference ImageDataFrame.scala: a dataframe with
columns label/data plus optional id/channels/height/width/encoded, with
`dataframe_column_select` projection support.

CoSDataFrameSource — reference DataFrameSource.scala: N typed tops
(STRING/INT/FLOAT/INT_ARRAY/FLOAT_ARRAY/RAW_IMAGE/ENCODED_IMAGE/
ENCODED_IMAGE_WITH_DIM), per-top transform_param, and `transpose` for
time-major 1-D tops (the LRCN caption path).
"""

from __future__ import annotations

import glob
import os
from typing import Iterator, List, Optional

import numpy as np
import torch

from ..proto import caffe_pb
from .image_source import ImageDataSource, ImageSample
from .source import STOP_MARK, DataSource
from .transformer import DataTransformer, decode_image


def _read_parquet_rows(path: str, columns=None) -> Iterator[dict]:
    import pyarrow.parquet as pq

    from ..utils.fsio import ensure_local
    path = ensure_local(path)
    files = [path]
    if os.path.isdir(path):
        files = sorted(glob.glob(os.path.join(path, "*.parquet"))) or \
            sorted(f for f in glob.glob(os.path.join(path, "*"))
                   if not os.path.basename(f).startswith((".", "_")))
    for f in files:
        table = pq.read_table(f, columns=columns)
        for batch in table.to_batches():
            cols = {name: batch.column(i)
                    for i, name in enumerate(batch.schema.names)}
            for r in range(batch.num_rows):
                yield {name: col[r].as_py() for name, col in cols.items()}


class ImageDataFrameSource(ImageDataSource):
    def init(self) -> None:
        p = self.layer_param.memory_data_param
        self.columns = list(p.dataframe_column_select) or None

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[ImageSample]:
        return self.persisted_epochs(lambda: self._epoch(rank, world),
                                     epochs)

    def _epoch(self, rank: int, world: int) -> Iterator[ImageSample]:
        for i, row in enumerate(_read_parquet_rows(self.source_path,
                                                   self.columns)):
            if i % world != rank:
                continue
            yield ImageSample(
                str(row.get("id", i)), float(row.get("label", 0.0)),
                int(row.get("channels", self.channels)),
                int(row.get("height", self.height)),
                int(row.get("width", self.width)),
                bool(row.get("encoded",
                             self.layer_param.memory_data_param
                             .image_encoded)),
                bytes(row["data"]))


class CoSDataFrameSource(DataSource):
    def __init__(self, conf, layer_param, is_train):
        super().__init__(conf, layer_param, is_train)
        p = layer_param.cos_data_param
        self._batch = int(p.batch_size)
        self.source_path = p.source
        self.tops = list(p.top)
        phase = caffe_pb.Phase.TRAIN if is_train else caffe_pb.Phase.TEST
        self.transformers = [
            DataTransformer(t.transform_param, phase)
            if t.has_field("transform_param") else None
            for t in self.tops]

    @property
    def batch_size(self) -> int:
        return self._batch

    def init(self) -> None:
        pass

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[dict]:
        return self.persisted_epochs(lambda: self._epoch(rank, world),
                                     epochs)

    def _epoch(self, rank: int, world: int) -> Iterator[dict]:
        for i, row in enumerate(_read_parquet_rows(self.source_path)):
            if i % world == rank:
                yield row

    def _fill_top(self, i: int, cfg, rows: List[dict]) -> torch.Tensor:
        T = caffe_pb.CoSTopType
        vals = [r.get(cfg.name) for r in rows]
        n = len(rows)
        if cfg.type in (T.INT, T.FLOAT, T.STRING):
            arr = np.asarray([float(v or 0) for v in vals], dtype=np.float32)
            return torch.from_numpy(arr.reshape(n, 1))
        if cfg.type in (T.INT_ARRAY, T.FLOAT_ARRAY):
            width = int(cfg.channels)
            arr = np.zeros((n, width), dtype=np.float32)
            for j, v in enumerate(vals):
                if v is None:
                    continue
                v = list(v)[:width]
                arr[j, :len(v)] = v
            t = torch.from_numpy(arr)
            if cfg.transpose:
                t = t.t().contiguous()  # time-major [T, N]
            return t
        # image types
        xf = self.transformers[i]
        imgs = []
        for v in vals:
            raw = bytes(v)
            if cfg.type == T.RAW_IMAGE:
                img = np.frombuffer(raw, dtype=np.uint8).reshape(
                    int(cfg.channels), int(cfg.height),
                    int(cfg.width)).transpose(1, 2, 0)
            else:
                resize = None
                if cfg.type == T.ENCODED_IMAGE_WITH_DIM and cfg.height:
                    resize = (int(cfg.height), int(cfg.width))
                img = decode_image(raw, color=int(cfg.channels) == 3,
                                   resize_hw=resize)
            imgs.append(img)
        if xf is not None:
            return xf.transform(imgs)
        return torch.from_numpy(
            np.stack([im.transpose(2, 0, 1) for im in imgs]).astype(
                np.float32))

    def next_batch(self, device, dtype) -> Optional[List[torch.Tensor]]:
        if getattr(self, "_drained", False):
            return None
        rows = []
        while len(rows) < self._batch:
            item = self.take()
            if item is STOP_MARK:
                if not rows:
                    return None
                self._drained = True
                break
            rows.append(item)
        out = []
        T = caffe_pb.CoSTopType
        for i, cfg in enumerate(self.tops):
            t = self._fill_top(i, cfg, rows)
            is_float_top = cfg.type in (T.RAW_IMAGE, T.ENCODED_IMAGE,
                                        T.ENCODED_IMAGE_WITH_DIM, T.FLOAT,
                                        T.FLOAT_ARRAY)
            out.append(t.to(device, dtype if is_float_top else None))
        return out
