"""Layer base class + registry.

Equivalent surface to `caffe::Layer<Dtype>` + `REGISTER_LAYER_CLASS`
(consumed in the reference by cos_data_layer.cpp:81 and the layer census of
SURVEY.md §3.6), re-designed for torch tensors and explicit fp32-master /
bf16-compute dtype handling.
"""

from __future__ import annotations

from typing import Dict, List, Type

import torch

from ...proto import caffe_pb
from ..blob import Blob

LAYER_REGISTRY: Dict[str, Type["Layer"]] = {}


def register_layer(name: str):
    def deco(cls):
        LAYER_REGISTRY[name] = cls
        cls.TYPE = name
        return cls
    return deco


def create_layer(param: caffe_pb.LayerParameter, net) -> "Layer":
    cls = LAYER_REGISTRY.get(param.type)
    if cls is None:
        raise ValueError(f"unknown layer type {param.type!r} (layer {param.name!r})")
    return cls(param, net)


class Layer:
    TYPE = ""

    def __init__(self, param: caffe_pb.LayerParameter, net):
        self.param = param
        self.name = param.name
        self.net = net
        self.phase = net.phase
        self.blobs: List[Blob] = []      # learnable params, fp32 master copies
        self._inplace: List[bool] = []

    # dtype the layer computes in (activations)
    @property
    def dtype(self) -> torch.dtype:
        return self.net.dtype

    @property
    def device(self) -> torch.device:
        return self.net.device

    def param_spec(self, i: int) -> caffe_pb.ParamSpec:
        specs = self.param.param
        return specs[i] if i < len(specs) else caffe_pb.ParamSpec()

    def add_param(self, shape, filler: caffe_pb.FillerParameter = None,
                  name: str = "") -> Blob:
        from .. import fillers
        spec = self.param_spec(len(self.blobs))
        share_name = spec.name
        if share_name and share_name in self.net.shared_params:
            # caffe named-param sharing (param { name: "w" } on several
            # layers aliases ONE blob: siamese nets, unrolled recurrences)
            b = self.net.shared_params[share_name]
            if list(b.shape) != [int(d) for d in shape]:
                raise ValueError(
                    f"shared param {share_name!r}: shape {list(b.shape)} "
                    f"!= {list(shape)} requested by layer {self.name!r}")
            self.blobs.append(b)
            return b
        b = Blob(shape, name=name or f"{self.name}_p{len(self.blobs)}",
                 dtype=torch.float32, device=self.device, alloc_diff=True)
        if filler is not None:
            fillers.fill(b.data, filler, generator=self.net.generator)
        b._lr_mult = spec.lr_mult
        b._decay_mult = spec.decay_mult
        if share_name:
            self.net.shared_params[share_name] = b
        self.blobs.append(b)
        return b

    # -- lifecycle -----------------------------------------------------------
    def setup(self, bottom: List[Blob], top: List[Blob]) -> None:
        pass

    def reshape(self, bottom: List[Blob], top: List[Blob]) -> None:
        pass

    def forward(self, bottom: List[Blob], top: List[Blob]) -> float:
        """Returns this layer's loss contribution (0 for non-loss layers)."""
        raise NotImplementedError

    def backward(self, top: List[Blob], propagate_down: List[bool],
                 bottom: List[Blob]) -> None:
        if any(propagate_down):
            raise NotImplementedError(
                f"{type(self).__name__} has no backward")

    # -- helpers -------------------------------------------------------------
    def cast(self, t: torch.Tensor) -> torch.Tensor:
        if t.dtype == self.dtype:
            return t
        if self.dtype == torch.bfloat16:
            # solver-managed params carry a bf16 shadow-arena view
            # (refreshed once per step): returning IT keeps the weight
            # tensor identity stable across steps, so per-weight caches
            # (packed GEMM layouts, fused repack registration) hit —
            # a fresh .to() copy per forward defeated them all
            sh = getattr(t, "_cos_bf16", None)
            if sh is not None:
                return sh
        return t.to(self.dtype)

    def weight(self, i: int = 0) -> torch.Tensor:
        """Param i's data in compute dtype."""
        return self.cast(self.blobs[i].data)

    def acc_param_diff(self, i: int, grad: torch.Tensor) -> None:
        b = self.blobs[i]
        d = b.ensure_diff()
        if getattr(b, "_grad_virgin", False):
            # first write this step: overwrite — Net.zero_param_diffs
            # marks instead of zeroing, saving a full pass over the
            # gradient arena every iteration
            d.copy_(grad.to(d.dtype))
            b._grad_virgin = False
        else:
            d.add_(grad.to(d.dtype))

    def _bias_arena(self):
        """The bias param's pre-zeroed fp32 arena diff view, when backward
        kernels may atomic-accumulate into it directly (flag set by
        Net.zero_param_diffs for 1-D params each step)."""
        if len(self.blobs) < 2:
            return None
        b = self.blobs[1]
        if getattr(b, "_diff_prezeroed", False):
            d = b.diff
            if d is not None and d.dtype == torch.float32 \
                    and d.is_contiguous():
                return d
        return None

    @staticmethod
    def acc_blob_diff(blob: Blob, dx: torch.Tensor, inplace: bool) -> None:
        """Accumulate dx into blob.diff (replace when the layer ran in-place:
        top and bottom share the tensor, so += would double-count)."""
        if inplace:
            blob.diff = dx
            return
        d = blob.diff
        if d is None or list(d.shape) != list(dx.shape):
            # first consumer: adopt the gradient tensor outright — the
            # zeros_like + add_ alternative costs two extra full passes
            # over the activation per backward layer
            blob.diff = dx
            return
        if d.dtype != dx.dtype:
            blob.diff = d.to(dx.dtype)
            d = blob.diff
        d.add_(dx)

    def loss_weight(self, i: int) -> float:
        lw = self.param.loss_weight
        if i < len(lw):
            return lw[i]
        if "Loss" in self.param.type and i == 0:
            return 1.0
        return 0.0

    @property
    def has_loss(self) -> bool:
        return any(self.loss_weight(i) != 0.0
                   for i in range(max(1, len(self.param.top))))

    def auto_top_blobs(self) -> int:
        """Min number of top blobs (used by Net to allocate)."""
        return len(self.param.top)
