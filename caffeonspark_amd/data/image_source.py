"""Image source base for MemoryData layers.

Mirrors ImageDataSource.scala: sample tuple
(id, label, channels, height, width, encoded, bytes); next_batch decodes /
resizes / transforms into the (data, label) tensor pair that is fed
zero-copy to the MemoryData layer.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..proto import caffe_pb
from .source import STOP_MARK, DataSource
from .transformer import DataTransformer, decode_image


class ImageSample:
    __slots__ = ("id", "label", "channels", "height", "width", "encoded",
                 "data")

    def __init__(self, id, label, channels, height, width, encoded, data):
        self.id = id
        self.label = float(label)
        self.channels = channels
        self.height = height
        self.width = width
        self.encoded = encoded
        self.data = data


class ImageDataSource(DataSource):
    def __init__(self, conf, layer_param, is_train):
        super().__init__(conf, layer_param, is_train)
        p = layer_param.memory_data_param
        self._batch = int(p.batch_size)
        self.channels = int(p.channels)
        self.height = int(p.height)
        self.width = int(p.width)
        self.source_path = p.source
        phase = caffe_pb.Phase.TRAIN if is_train else caffe_pb.Phase.TEST
        self.transformer = DataTransformer(
            layer_param.transform_param, phase,
            seed=getattr(conf, "seed", None))

    @property
    def batch_size(self) -> int:
        return self._batch

    def _decode(self, s: ImageSample) -> np.ndarray:
        want_resize = getattr(self.conf, "resize", False)
        if s.encoded:
            resize = None
            if self.height and self.width and (
                    want_resize or not self.transformer.param.crop_size):
                resize = (self.height, self.width)
            return decode_image(s.data, color=(self.channels == 3),
                                resize_hw=resize)
        # raw Datum bytes are CHW (caffe convention); reorder to HWC like
        # the reference does for cv::Mat (LmdbRDD.scala:270-281)
        c = s.channels or self.channels
        h = s.height or self.height
        w = s.width or self.width
        dt = np.float32 if getattr(s, "is_float", False) else np.uint8
        arr = np.frombuffer(s.data, dtype=dt)
        return arr.reshape(c, h, w).transpose(1, 2, 0)

    def next_batch(self, device, dtype) -> Optional[List[torch.Tensor]]:
        if getattr(self, "_drained", False):
            return None
        images, labels = [], []
        while len(images) < self._batch:
            item = self.take()
            if item is STOP_MARK:
                if not images:
                    return None
                # partial final batch consumed the stop mark: remember so
                # the NEXT call terminates instead of blocking forever
                self._drained = True
                break
            images.append(self._decode(item))
            labels.append(item.label)
        data = self.transformer.transform(images)
        label = torch.tensor(labels, dtype=torch.float32)
        return [data.to(device, dtype), label.to(device)]
