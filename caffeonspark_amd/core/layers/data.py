"""Data-entry layers: MemoryData, CoSData, DummyData, Input.

MemoryData mirrors the reference's zero-copy feed path (reference
MemoryInputAdapter.cpp:24-33 calls `MemoryDataLayer::Reset(data,label,n)`
with a GPU pointer); CoSData mirrors cos_data_layer.cpp:51-78 (`Reset`
with one device pointer per top).  Here "pointer adoption" is simply
adopting the fed torch tensors as the tops' data.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from .. import fillers
from .base import Layer, register_layer


@register_layer("MemoryData")
class MemoryDataLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.memory_data_param
        self.batch_size = int(p.batch_size)
        self.channels = int(p.channels)
        self.height = int(p.height)
        self.width = int(p.width)
        self._data: Optional[torch.Tensor] = None
        self._label: Optional[torch.Tensor] = None

    def reset(self, data: torch.Tensor, label: torch.Tensor, n: int = -1):
        """Zero-copy adoption of a prepared batch (reference: Reset())."""
        self._data = data
        self._label = label

    def forward(self, bottom, top):
        if self._data is None:
            raise RuntimeError(
                f"MemoryData layer {self.name!r}: no batch fed (call reset())")
        top[0].data = self._data
        if len(top) > 1:
            top[1].data = self._label
        return 0.0

    def backward(self, top, propagate_down, bottom):
        pass


@register_layer("CoSData")
class CoSDataLayer(Layer):
    """N typed tops with per-top shape / sample_num_axes / transpose
    (time-major) semantics, fed zero-copy by the data pipeline."""

    def setup(self, bottom, top):
        p = self.param.cos_data_param
        self.batch_size = int(p.batch_size)
        self.tops_cfg = list(p.top)
        if len(self.tops_cfg) != len(self.param.top):
            raise ValueError(
                f"CoSData {self.name!r}: {len(self.param.top)} tops declared "
                f"but {len(self.tops_cfg)} cos_data_param.top entries")
        self._fed: Optional[List[torch.Tensor]] = None

    def top_shape(self, i: int, batch: int) -> List[int]:
        cfg = self.tops_cfg[i]
        c = int(cfg.out_channels or cfg.channels)
        h = int(cfg.out_height or cfg.height)
        w = int(cfg.out_width or cfg.width)
        dims = [c, h, w][:int(cfg.sample_num_axes)]
        if cfg.transpose:
            # time-major: sample axis 0 becomes time, batch second
            assert int(cfg.sample_num_axes) <= 1
            return dims + [batch]
        return [batch] + dims

    def reset(self, tensors: List[torch.Tensor]):
        self._fed = tensors

    def forward(self, bottom, top):
        if self._fed is None:
            raise RuntimeError(
                f"CoSData layer {self.name!r}: no batch fed (call reset())")
        for t, tensor in zip(top, self._fed):
            t.data = tensor
        return 0.0

    def backward(self, top, propagate_down, bottom):
        pass


@register_layer("DummyData")
class DummyDataLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.dummy_data_param
        self.shapes = []
        if p.shape:
            self.shapes = [[int(d) for d in s.dim] for s in p.shape]
        else:
            for i in range(len(self.param.top)):
                def pick(rep):
                    return int(rep[i]) if i < len(rep) else int(rep[0])
                self.shapes.append([pick(p.num), pick(p.channels),
                                    pick(p.height), pick(p.width)])
        while len(self.shapes) < len(self.param.top):
            self.shapes.append(self.shapes[-1])
        self.fillers_ = list(p.data_filler)

    def forward(self, bottom, top):
        for i, t in enumerate(top):
            if list(t.data.shape) != self.shapes[i] or \
                    t.data.device != self.device:
                t.data = torch.zeros(self.shapes[i], dtype=self.dtype,
                                     device=self.device)
                f = self.fillers_[i] if i < len(self.fillers_) else \
                    (self.fillers_[0] if self.fillers_ else None)
                if f is not None:
                    fillers.fill(t.data, f, generator=self.net.generator)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        pass


@register_layer("Input")
class InputLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.input_param
        self.shapes = [[int(d) for d in s.dim] for s in p.shape]
        while len(self.shapes) < len(self.param.top):
            self.shapes.append(self.shapes[-1])

    def forward(self, bottom, top):
        for i, t in enumerate(top):
            if t.data.numel() == 0:
                t.data = torch.zeros(self.shapes[i], dtype=self.dtype,
                                     device=self.device)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        pass
