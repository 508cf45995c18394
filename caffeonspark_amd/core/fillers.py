"""Weight fillers (constant / uniform / gaussian / xavier / msra / ...).

Semantics follow upstream Caffe's filler.hpp (exercised by every
weight_filler/bias_filler entry in the reference's data/*.prototxt).
"""

from __future__ import annotations

import math

import torch

from ..proto import caffe_pb


def fill(tensor: torch.Tensor, param: caffe_pb.FillerParameter,
         generator: torch.Generator = None) -> None:
    if tensor.device.type != "cpu" or tensor.dtype != torch.float32:
        # always draw in fp32 on CPU (identical streams across devices AND
        # dtypes — normal_ on a bf16 tensor draws a different sequence),
        # then cast
        tmp = torch.empty(tensor.shape, dtype=torch.float32)
        fill(tmp, param, generator)
        tensor.copy_(tmp.to(tensor.dtype))
        return
    ftype = param.type or "constant"
    if ftype == "constant":
        tensor.fill_(param.value)
    elif ftype == "uniform":
        lo, hi = param.min, param.max
        tensor.uniform_(lo, hi, generator=generator)
    elif ftype == "gaussian":
        tensor.normal_(param.mean, param.std, generator=generator)
        # sparse gaussian (param.sparse >= 0) not supported; no reference
        # config uses it
    elif ftype in ("xavier", "msra"):
        # fan computation: blob shape [out, in, kh, kw] or [out, in]
        shape = tensor.shape
        fan_out = shape[0] if len(shape) > 0 else 1
        fan_in = int(tensor.numel() // max(1, fan_out))
        vn = param.variance_norm
        if vn == caffe_pb.VarianceNorm.FAN_IN:
            n = fan_in
        elif vn == caffe_pb.VarianceNorm.FAN_OUT:
            n = fan_out
        else:
            n = (fan_in + fan_out) / 2.0 if vn == caffe_pb.VarianceNorm.AVERAGE else fan_in
        if ftype == "xavier":
            scale = math.sqrt(3.0 / n)
            tensor.uniform_(-scale, scale, generator=generator)
        else:  # msra
            std = math.sqrt(2.0 / n)
            tensor.normal_(0.0, std, generator=generator)
    elif ftype == "positive_unitball":
        tensor.uniform_(0, 1, generator=generator)
        flat = tensor.view(tensor.shape[0], -1)
        flat /= flat.sum(dim=1, keepdim=True)
    elif ftype == "bilinear":
        # upsampling deconv filter
        shape = tensor.shape
        f = math.ceil(shape[-1] / 2.0)
        c = (2 * f - 1 - f % 2) / (2.0 * f)
        for i in range(tensor.numel()):
            x = i % shape[-1]
            y = (i // shape[-1]) % shape[-2]
            tensor.view(-1)[i] = (1 - abs(x / f - c)) * (1 - abs(y / f - c))
    else:
        raise ValueError(f"unknown filler type {ftype!r}")
