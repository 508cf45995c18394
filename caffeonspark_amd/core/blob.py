"""Blob: named tensor pair (data, diff) backing every layer input/output.

Mirrors the role of `caffe::Blob<float>` + SyncedMemory in the reference
(consumed via JNI in reference caffe-distri/src/main/cpp/jni/JniFloatBlob.cpp)
but is a thin wrapper over torch tensors: device residency, dtype and
zero-copy external adoption (`set_data`, the reference's `set_gpu_data`
path — MemoryInputAdapter.cpp:24-33) come from torch/HIP.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from ..proto import caffe_pb


class Blob:
    __slots__ = ("name", "data", "diff", "_lr_mult", "_decay_mult",
                 "_loss_weight", "_grad_virgin", "_diff_prezeroed")

    def __init__(self, shape: Sequence[int] = (), *, name: str = "",
                 dtype: torch.dtype = torch.float32,
                 device: Optional[torch.device] = None,
                 alloc_diff: bool = False):
        self.name = name
        device = device or torch.device("cpu")
        self.data = torch.zeros(list(shape), dtype=dtype, device=device)
        self.diff = (torch.zeros(list(shape), dtype=dtype, device=device)
                     if alloc_diff else None)
        self._lr_mult = 1.0
        self._decay_mult = 1.0

    # -- shape ---------------------------------------------------------------
    @property
    def shape(self) -> List[int]:
        return list(self.data.shape)

    @property
    def count(self) -> int:
        return self.data.numel()

    def reshape(self, shape: Sequence[int]) -> "Blob":
        if list(self.data.shape) != list(shape):
            self.data = torch.zeros(list(shape), dtype=self.data.dtype,
                                    device=self.data.device)
            if self.diff is not None:
                self.diff = torch.zeros_like(self.data)
        return self

    # -- zero-copy adoption (reference: Blob::set_cpu_data / set_gpu_data) ---
    def set_data(self, tensor: torch.Tensor) -> None:
        self.data = tensor

    def ensure_diff(self) -> torch.Tensor:
        if self.diff is None or self.diff.shape != self.data.shape \
                or self.diff.device != self.data.device:
            self.diff = torch.zeros_like(self.data)
        return self.diff

    def zero_diff(self) -> None:
        if self.diff is not None:
            self.diff.zero_()

    def to_(self, device: torch.device) -> "Blob":
        self.data = self.data.to(device)
        if self.diff is not None:
            self.diff = self.diff.to(device)
        return self

    # -- proto IO ------------------------------------------------------------
    def to_proto(self, *, write_diff: bool = False, raw: bool = True) -> caffe_pb.BlobProto:
        """Serialize. raw=True stores little-endian bytes in our extension
        field (fast, exact); the legacy repeated-float field is always
        written for float32 blobs so upstream Caffe can read the file."""
        proto = caffe_pb.BlobProto()
        proto.shape = caffe_pb.BlobShape(dim=[int(d) for d in self.data.shape])
        t = self.data.detach().cpu().contiguous()
        if t.dtype == torch.float32:
            proto.data = t.view(-1).numpy()  # codec fast-paths ndarrays
        else:
            proto.raw_data = t.numpy().tobytes()
            proto.raw_dtype = str(t.dtype).replace("torch.", "")
        if write_diff and self.diff is not None:
            proto.diff = self.diff.detach().cpu().float().view(-1).tolist()
        return proto

    @staticmethod
    def _tensor_to_proto(t: torch.Tensor) -> caffe_pb.BlobProto:
        proto = caffe_pb.BlobProto()
        proto.shape = caffe_pb.BlobShape(dim=[int(d) for d in t.shape])
        proto.data = t.detach().cpu().float().view(-1).numpy()
        return proto

    @staticmethod
    def shape_from_proto(proto: caffe_pb.BlobProto) -> List[int]:
        if proto.has_field("shape"):
            return [int(d) for d in proto.shape.dim]
        dims = [proto.num, proto.channels, proto.height, proto.width]
        # legacy blobs always carry 4 dims
        return [int(d) for d in dims]

    def from_proto(self, proto: caffe_pb.BlobProto, *, reshape: bool = True) -> "Blob":
        shape = Blob.shape_from_proto(proto)
        if reshape:
            self.reshape(shape)
        device, dtype = self.data.device, self.data.dtype
        if proto.has_field("raw_data"):
            import numpy as np
            np_dtype = {"float32": np.float32, "bfloat16": np.uint16,
                        "float16": np.float16, "int64": np.int64}.get(
                            proto.raw_dtype, np.float32)
            arr = np.frombuffer(proto.raw_data, dtype=np_dtype).copy()
            t = torch.from_numpy(arr)
            if proto.raw_dtype == "bfloat16":
                t = t.view(torch.bfloat16)
            src = t.reshape(shape)
        elif proto.has_field("double_data"):
            import numpy as np
            src = torch.from_numpy(
                np.asarray(proto.double_data, dtype=np.float64).copy()
            ).reshape(shape)
        else:
            import numpy as np
            src = torch.from_numpy(
                np.asarray(proto.data, dtype=np.float32).copy()
            ).reshape(shape)
        src = src.to(device=device, dtype=dtype)
        if list(self.data.shape) == shape:
            # copy IN PLACE: param blobs are views of the solver's flat
            # arena — reassignment would silently detach them from the
            # optimizer/all-reduce/bf16-shadow paths
            self.data.copy_(src)
        else:
            self.data = src
        if proto.has_field("diff"):
            d = torch.tensor(proto.diff, dtype=torch.float32).reshape(
                shape).to(device=device, dtype=dtype)
            if self.diff is not None and list(self.diff.shape) == shape:
                self.diff.copy_(d)
            else:
                self.diff = d
        return self
