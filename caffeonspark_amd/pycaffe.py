"""pycaffe-style inference API (reference: `import caffe` used by
examples/ImageCaption.py through the caffe-public python bindings).

Provides the small surface those scripts rely on:

    import caffeonspark_amd.pycaffe as caffe
    net = caffe.Net("deploy.prototxt", "model.caffemodel", caffe.TEST)
    net.blobs["data"].data[...] = batch
    out = net.forward()
    probs = net.blobs["prob"].data
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from .core import net_from_prototxt
from .proto import caffe_pb, read_binary_proto

TRAIN = caffe_pb.Phase.TRAIN
TEST = caffe_pb.Phase.TEST


class _BlobView:
    """numpy-backed view of a Blob (pycaffe's blob.data semantics)."""

    def __init__(self, blob):
        self._blob = blob

    @property
    def data(self) -> np.ndarray:
        t = self._blob.data
        arr = t.detach().float().cpu().numpy()
        self._cache = arr
        return arr

    @data.setter
    def data(self, value):
        self.set_data(value)

    def set_data(self, value) -> None:
        t = torch.as_tensor(np.ascontiguousarray(value))
        self._blob.data = t.to(self._blob.data.device,
                               self._blob.data.dtype)

    @property
    def shape(self):
        return tuple(self._blob.shape)


class Net:
    def __init__(self, proto_file: str, weights: Optional[str] = None,
                 phase: int = TEST, *, stages: Optional[List[str]] = None,
                 device: Optional[torch.device] = None):
        state = caffe_pb.NetState(phase=phase, stage=list(stages or []))
        dtype = torch.bfloat16 if (device is not None and
                                   device.type == "cuda") else torch.float32
        self._net = net_from_prototxt(proto_file, state=state,
                                      device=device, dtype=dtype)
        if weights:
            if weights.endswith(".h5"):
                from .utils import hdf5
                hdf5.load_net(weights, self._net)
            else:
                self._net.copy_trained_layers_from(
                    read_binary_proto(weights, caffe_pb.NetParameter))
        self.blobs: Dict[str, _BlobView] = {
            name: _BlobView(b) for name, b in self._net.blob_map.items()}
        self.params: Dict[str, List[_BlobView]] = {
            l.name: [_BlobView(b) for b in l.blobs]
            for l in self._net.layers if l.blobs}

    def forward(self, **inputs) -> Dict[str, np.ndarray]:
        for name, value in inputs.items():
            self.blobs[name].set_data(value)
        self._net.forward()
        return {name: self.blobs[name].data
                for name in self._net.output_blob_names()}

    def backward(self) -> Dict[str, np.ndarray]:
        """pycaffe-style backward: populate blob diffs from the loss tops
        and return input-blob gradients."""
        self._net.backward()
        out = {}
        for name, b in self._net.blob_map.items():
            if b.diff is not None:
                out[name] = b.diff.detach().float().cpu().numpy()
        return out

    def save(self, path: str) -> None:
        """Write learned weights as a .caffemodel (binary NetParameter)."""
        from .proto import write_binary_proto
        write_binary_proto(path, self._net.to_proto())

    @property
    def layer_names(self) -> List[str]:
        return [l.name for l in self._net.layers]

    @property
    def outputs(self) -> List[str]:
        return list(self._net.output_blob_names())
