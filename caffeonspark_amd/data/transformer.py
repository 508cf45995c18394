"""DataTransformer: scale / mean-file / mean-value / crop / mirror.

Equivalent of `caffe::DataTransformer<float>` as driven from the JVM in the
reference (JniFloatDataTransformer.cpp:99-104 calls
`Transform(vector<cv::Mat>&, Blob*)`): runs CPU-side in the reader threads,
producing the batch tensor that is fed zero-copy to the data layer.
Image decode uses PIL (the reference used cv::imdecode through a JNI Mat
wrapper).
"""

from __future__ import annotations

import io
from typing import Optional, Sequence

import numpy as np
import torch

from ..proto import caffe_pb


def decode_image(data: bytes, *, color: Optional[bool] = None,
                 resize_hw: Optional[tuple] = None) -> np.ndarray:
    """Decode an encoded image to HWC uint8 BGR (caffe channel order)."""
    from PIL import Image
    img = Image.open(io.BytesIO(data))
    if color is None:
        color = img.mode not in ("L", "1", "I;16")
    img = img.convert("RGB" if color else "L")
    if resize_hw is not None:
        h, w = resize_hw
        img = img.resize((w, h))
    arr = np.asarray(img)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    elif arr.shape[2] == 3:
        arr = arr[:, :, ::-1]  # RGB -> BGR like cv::imdecode
    return arr


class DataTransformer:
    def __init__(self, param: caffe_pb.TransformationParameter,
                 phase: int = caffe_pb.Phase.TRAIN,
                 seed: Optional[int] = None):
        self.param = param
        self.phase = phase
        self.rng = np.random.RandomState(seed)
        self.mean: Optional[np.ndarray] = None
        if param.mean_file:
            bp = caffe_pb.BlobProto.FromString(
                open(param.mean_file, "rb").read())
            shape = [d for d in (bp.shape.dim if bp.has_field("shape") else
                                 [bp.channels, bp.height, bp.width]) if d]
            self.mean = np.asarray(bp.data, dtype=np.float32).reshape(shape)
            if self.mean.ndim == 4:
                self.mean = self.mean[0]
        elif param.mean_value:
            self.mean = np.asarray(list(param.mean_value),
                                   dtype=np.float32).reshape(-1, 1, 1)

    def output_shape(self, c: int, h: int, w: int):
        crop = int(self.param.crop_size)
        if crop:
            return c, crop, crop
        return c, h, w

    def transform_one(self, img_hwc: np.ndarray) -> np.ndarray:
        """uint8/float HWC -> float32 CHW transformed."""
        arr = img_hwc.astype(np.float32).transpose(2, 0, 1)  # CHW
        crop = int(self.param.crop_size)
        c, h, w = arr.shape
        if self.mean is not None:
            if self.mean.shape[-1] == 1 or self.mean.shape == (c, 1, 1):
                arr = arr - self.mean.reshape(c, 1, 1)
            else:
                arr = arr - self.mean
        if crop and (h > crop or w > crop):
            if self.phase == caffe_pb.Phase.TRAIN:
                h_off = self.rng.randint(0, h - crop + 1)
                w_off = self.rng.randint(0, w - crop + 1)
            else:
                h_off = (h - crop) // 2
                w_off = (w - crop) // 2
            arr = arr[:, h_off:h_off + crop, w_off:w_off + crop]
        if self.param.mirror and self.phase == caffe_pb.Phase.TRAIN \
                and self.rng.randint(2):
            arr = arr[:, :, ::-1]
        if self.param.scale != 1.0:
            arr = arr * self.param.scale
        return np.ascontiguousarray(arr)

    def transform(self, images: Sequence[np.ndarray],
                  out: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Batch of HWC arrays -> [N,C,H,W] float32 tensor."""
        batch = np.stack([self.transform_one(im) for im in images])
        t = torch.from_numpy(batch)
        if out is not None:
            out.copy_(t)
            return out
        return t
