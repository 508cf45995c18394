"""Net: graph builder + topological forward/backward over the layer catalog.

Re-implements the `caffe::Net` surface the reference consumes
(CaffeNet.cpp:38-97, 321-331, 634-696 — SURVEY.md §2.5): construction from
NetParameter with phase/stage/level filtering (NetStateRule), named-blob
access, weight sharing between train/test nets, CopyTrainedLayersFrom, and
loss accumulation via per-top loss weights.  Diff fan-in is accumulated
directly at the blob level instead of materializing Caffe's automatic
Split layers.
"""

from __future__ import annotations

import os

from typing import Dict, List, Optional

import torch

from .. import ops
from ..proto import caffe_pb, text_format
from .blob import Blob
from .layers import base as layer_base
from .layers.base import create_layer


def state_matches(rule: caffe_pb.NetStateRule, state: caffe_pb.NetState) -> bool:
    if rule.has_field("phase") and rule.phase != state.phase:
        return False
    if rule.has_field("min_level") and state.level < rule.min_level:
        return False
    if rule.has_field("max_level") and state.level > rule.max_level:
        return False
    stages = set(state.stage)
    for s in rule.stage:
        if s not in stages:
            return False
    for s in rule.not_stage:
        if s in stages:
            return False
    return True


def layer_included(lp: caffe_pb.LayerParameter, state: caffe_pb.NetState) -> bool:
    if lp.include:
        return any(state_matches(r, state) for r in lp.include)
    if lp.exclude:
        return not any(state_matches(r, state) for r in lp.exclude)
    return True


def filter_net(param: caffe_pb.NetParameter,
               state: caffe_pb.NetState) -> caffe_pb.NetParameter:
    out = caffe_pb.NetParameter()
    out.CopyFrom(param)
    out.layer = [lp for lp in out.layer if layer_included(lp, state)]
    return out


class Net:
    def __init__(self, param: caffe_pb.NetParameter,
                 state: Optional[caffe_pb.NetState] = None, *,
                 device: Optional[torch.device] = None,
                 dtype: torch.dtype = torch.float32,
                 seed: Optional[int] = None):
        self.state = state or caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN)
        self.phase = self.state.phase
        self.device = device or torch.device("cpu")
        self.dtype = dtype
        self.generator = torch.Generator(device="cpu")
        if seed is not None and seed >= 0:
            self.generator.manual_seed(int(seed))
        self.param = filter_net(param, self.state)
        self.name = self.param.name
        self.layers: List[layer_base.Layer] = []
        self.blob_map: Dict[str, Blob] = {}
        self.layer_bottoms: List[List[Blob]] = []
        self.layer_tops: List[List[Blob]] = []
        self.layer_need_backward: List[bool] = []
        self.layer_prop_down: List[List[bool]] = []
        self.shared_params: Dict[str, Blob] = {}
        self._loss_tops: List[tuple] = []  # (layer_idx, top_idx, weight)
        # ops-dispatch mode this net declares while its forward/backward
        # runs: fp32-on-GPU routes activation ops to the reference torch
        # impls (the declared fp32 path), bf16 forces the native kernels
        if self.device.type == "cuda":
            self._ops_mode = "fp32" if dtype == torch.float32 else "bf16"
        else:
            self._ops_mode = None
        self._build()

    # ------------------------------------------------------------------ build
    def _build(self) -> None:
        # handle legacy `input:` fields as an implicit Input layer
        if self.param.input:
            shapes = []
            if self.param.input_shape:
                shapes = [[int(d) for d in s.dim] for s in self.param.input_shape]
            elif self.param.input_dim:
                dims = [int(d) for d in self.param.input_dim]
                shapes = [dims[i:i + 4] for i in range(0, len(dims), 4)]
            for i, name in enumerate(self.param.input):
                blob = Blob(shapes[i] if i < len(shapes) else [1],
                            name=name, dtype=self.dtype, device=self.device)
                self.blob_map[name] = blob

        blob_needs_grad: Dict[str, bool] = {n: False for n in self.blob_map}

        for lp in self.param.layer:
            layer = create_layer(lp, self)
            bottoms = []
            for bname in lp.bottom:
                if bname not in self.blob_map:
                    raise ValueError(
                        f"layer {lp.name!r} needs unknown blob {bname!r}")
                bottoms.append(self.blob_map[bname])
            tops = []
            for tname in lp.top:
                if tname in lp.bottom:
                    tops.append(self.blob_map[tname])  # in-place
                else:
                    blob = Blob([0], name=tname, dtype=self.dtype,
                                device=self.device)
                    self.blob_map[tname] = blob
                    tops.append(blob)

            layer.setup(bottoms, tops)
            # data layers need a placeholder batch so the shape-propagation
            # forward below works
            self._prefeed_placeholder(layer)
            layer.forward(bottoms, tops)  # shape propagation

            prop = []
            for i, bname in enumerate(lp.bottom):
                p = blob_needs_grad.get(bname, False) or \
                    self.param.force_backward
                if i < len(lp.propagate_down):
                    p = p and lp.propagate_down[i]
                if lp.type in ("SoftmaxWithLoss", "Accuracy") and i >= 1:
                    p = False
                if lp.type == "Embed" and i == 0:
                    p = False
                if lp.type == "LSTM" and i == 1:
                    p = False  # cont markers
                prop.append(p)
            has_params = any(b._lr_mult != 0 for b in layer.blobs)
            need_bw = (any(prop) or has_params) and lp.type not in (
                "Accuracy", "Silence")
            grad_out = need_bw and lp.type not in ("Accuracy",)
            for tname in lp.top:
                blob_needs_grad[tname] = grad_out

            idx = len(self.layers)
            self.layers.append(layer)
            self.layer_bottoms.append(bottoms)
            self.layer_tops.append(tops)
            self.layer_need_backward.append(need_bw)
            self.layer_prop_down.append(prop)
            for ti in range(len(lp.top)):
                w = layer.loss_weight(ti)
                if w != 0.0:
                    self._loss_tops.append((idx, ti, w))
        self._fuse_relu_peephole()
        self._fuse_concat_peephole()

    def _fuse_relu_peephole(self) -> None:
        """Fuse Conv/IP + in-place ReLU pairs on GPU: the GEMM epilogue
        applies the rectification, the ReLU layer's forward becomes a
        no-op (backward is unchanged — it masks by top data, which the
        fused producer already rectified)."""
        if self.device.type != "cuda":
            return
        for a, b in zip(self.layers, self.layers[1:]):
            if a.param.type not in ("Convolution", "InnerProduct"):
                continue
            if b.param.type != "ReLU" or getattr(b, "slope", 0.0) != 0.0:
                continue
            if list(b.param.bottom) != list(b.param.top):
                continue
            if not a.param.top or b.param.bottom[0] != a.param.top[0]:
                continue
            a._fuse_relu = True
            b._fused_upstream = True
            if len(a.blobs) > 1 and a.blobs[1].data.dim() == 1:
                # backward fusion: the ReLU's gradient pass also column-
                # sums the producer's bias gradient into the arena
                b._db_producer = a

    def _fuse_concat_peephole(self) -> None:
        """Inception fusion (GPU): when every input of a channel Concat
        is a Convolution top consumed only by that concat (after the
        in-place-ReLU fusion), the branch convs write straight into
        channel windows of the concat's output buffer (GEMM ldc = total
        channels) and backward hands out channel-slice views — the
        concat's forward copies and backward slicing disappear."""
        if self.device.type != "cuda" or \
                not int(os.environ.get("COS_CONCAT_FUSE", "1")):
            return
        # consumer counts by blob name (in-place layers don't count:
        # they pass the blob through)
        consumers: dict = {}
        for lp in (l.param for l in self.layers):
            tops = set(lp.top)
            for bn in lp.bottom:
                if bn not in tops:
                    consumers[bn] = consumers.get(bn, 0) + 1
        producer = {}
        for l in self.layers:
            if l.param.type == "Convolution" and l.param.top:
                producer[l.param.top[0]] = l
        out_names = set(self.output_blob_names()) \
            if hasattr(self, "output_blob_names") else set()
        for l in self.layers:
            if l.param.type != "Concat" or getattr(l, "axis", None) != 1:
                continue
            convs = [producer.get(bn) for bn in l.param.bottom]
            if (any(c is None for c in convs)
                    or any(consumers.get(bn, 0) != 1 for bn in l.param.bottom)
                    or any(bn in out_names for bn in l.param.bottom)
                    or len(set(id(c) for c in convs)) != len(convs)):
                continue
            off = 0
            for c in convs:
                c._concat_out = (l, off)
                off += c.num_output
            l._fused = True
            l._fused_ctot = off
            l._fused_buf = None
            # dense-ReLU backward: when every branch ends in a fused
            # in-place ReLU, ONE dense relu_bwd over the whole concat
            # replaces per-branch strided-window passes (half their
            # cache-line utilization); branch ReLUs then pass through
            relus = {}
            for rl in self.layers:
                if rl.param.type == "ReLU" and rl.param.top and \
                        getattr(rl, "_fused_upstream", False):
                    relus[rl.param.top[0]] = rl
            br = [relus.get(bn) for bn in l.param.bottom]
            l._branch_relus = br if all(r is not None for r in br) else None

    def _prefeed_placeholder(self, layer) -> None:
        from .layers.data import CoSDataLayer, MemoryDataLayer
        lp = layer.param
        if isinstance(layer, MemoryDataLayer):
            p = lp.memory_data_param
            n = max(1, int(p.batch_size))
            data = torch.zeros(n, int(p.channels), int(p.height), int(p.width),
                               dtype=self.dtype, device=self.device)
            label = torch.zeros(n, dtype=self.dtype, device=self.device)
            layer.reset(data, label)
        elif isinstance(layer, CoSDataLayer):
            n = max(1, layer.batch_size)
            fed = []
            for i, cfg in enumerate(layer.tops_cfg):
                shape = layer.top_shape(i, n)
                dt = torch.float32 if cfg.type in (
                    caffe_pb.CoSTopType.INT, caffe_pb.CoSTopType.INT_ARRAY,
                    caffe_pb.CoSTopType.STRING) else self.dtype
                fed.append(torch.zeros(shape, dtype=dt, device=self.device))
            layer.reset(fed)

    # ---------------------------------------------------------------- forward
    def forward(self, read_loss: bool = True) -> float:
        """read_loss=False skips the device->host loss readout (the only
        sync in a forward), for hipGraph capture and sync-free stepping."""
        loss = 0.0
        refresh = getattr(self, "_bf16_refresh", None)
        if refresh is not None:
            refresh()      # re-cast the fp32 master arena to bf16 once
        ops.set_active_gpu_mode(self._ops_mode)
        try:
            for layer, bottoms, tops in zip(self.layers, self.layer_bottoms,
                                            self.layer_tops):
                layer.forward(bottoms, tops)
        finally:
            ops.set_active_gpu_mode(None)
        if read_loss:
            for (li, ti, w) in self._loss_tops:
                loss += w * float(self.layer_tops[li][ti].data.float().sum())
        return loss

    # --------------------------------------------------------------- backward
    def backward(self, on_layer_done=None) -> None:
        # clear activation diffs (param diffs persist for iter_size accum)
        for blob in self.blob_map.values():
            blob.diff = None
        # seed loss tops
        for (li, ti, w) in self._loss_tops:
            top = self.layer_tops[li][ti]
            top.diff = torch.full_like(top.data, w, dtype=torch.float32)
            top._loss_weight = w  # host-side copy: layers avoid a sync
        ops.set_active_gpu_mode(self._ops_mode)
        if self._ops_mode == "bf16":
            # reset the per-backward scratch sequence + one fused zero of
            # every split-K/colsum accumulator (replaces ~2 FillFunctor
            # launches per conv per step)
            from ..ops import gpu as _gops
            _gops.begin_backward()
        try:
            for i in range(len(self.layers) - 1, -1, -1):
                if not self.layer_need_backward[i]:
                    continue
                tops = self.layer_tops[i]
                if all(t.diff is None for t in tops) and not any(
                        b._lr_mult != 0 for b in self.layers[i].blobs):
                    continue
                for t in tops:
                    if t.diff is None:
                        t.ensure_diff()
                self.layers[i].backward(tops, self.layer_prop_down[i],
                                        self.layer_bottoms[i])
                if on_layer_done is not None and self.layers[i].blobs:
                    on_layer_done(self.layers[i])
        finally:
            ops.set_active_gpu_mode(None)

    def forward_backward(self) -> float:
        loss = self.forward()
        self.backward()
        return loss

    # ------------------------------------------------------------------ params
    def learnable_params(self) -> List[Blob]:
        seen, out = set(), []
        for layer in self.layers:
            for b in layer.blobs:
                if id(b) not in seen:
                    seen.add(id(b))
                    out.append(b)
        return out

    def zero_param_diffs(self) -> None:
        # 2D+ params: mark rather than zero — the first acc_param_diff
        # this step copies (or the GEMM plain-stores into the arena
        # slice) instead of accumulating.  1-D params (biases): ONE fused
        # zero of all their diffs, so backward kernels (colsum) can
        # atomic-add straight into the arena with no staging copy.
        vec = []
        for b in self.learnable_params():
            d = b.diff
            if b.data.dim() == 1 and d is not None \
                    and d.dtype == torch.float32 and d.is_contiguous():
                vec.append(d)
                b._grad_virgin = False
                b._diff_prezeroed = True
            else:
                b._grad_virgin = True
                b._diff_prezeroed = False
        if vec:
            torch._foreach_zero_(vec)

    # ----------------------------------------------------------------- access
    def blob_by_name(self, name: str) -> Blob:
        return self.blob_map[name]

    @property
    def blob_names(self) -> List[str]:
        return list(self.blob_map.keys())

    def output_blob_names(self) -> List[str]:
        consumed = set()
        for lp in self.param.layer:
            consumed.update(lp.bottom)
        outs = []
        for lp in self.param.layer:
            for t in lp.top:
                if t not in consumed and t not in outs:
                    outs.append(t)
        return outs

    def layer_by_name(self, name: str) -> Optional[layer_base.Layer]:
        for l in self.layers:
            if l.name == name:
                return l
        return None

    def data_layers(self):
        from .layers.data import CoSDataLayer, MemoryDataLayer
        return [l for l in self.layers
                if isinstance(l, (MemoryDataLayer, CoSDataLayer))]

    # ---------------------------------------------------------------- sharing
    def share_trained_layers_with(self, other: "Net") -> None:
        """Point this net's param blobs at `other`'s (reference:
        Net::ShareTrainedLayersWith, used by CaffeNet::validation)."""
        for layer in self.layers:
            src = other.layer_by_name(layer.name)
            if src is None:
                continue
            for i, b in enumerate(layer.blobs):
                if i < len(src.blobs):
                    b.data = src.blobs[i].data

    def copy_trained_layers_from(self, net_param: caffe_pb.NetParameter) -> None:
        by_name = {lp.name: lp for lp in net_param.layer}
        for layer in self.layers:
            lp = by_name.get(layer.name)
            if lp is None or not lp.blobs:
                continue
            for i, bp in enumerate(lp.blobs):
                if i >= len(layer.blobs):
                    break
                shape = Blob.shape_from_proto(bp)
                dst = layer.blobs[i]
                if int(torch.tensor(shape).prod()) != dst.count:
                    raise ValueError(
                        f"shape mismatch loading {layer.name} blob {i}: "
                        f"{shape} vs {dst.shape}")
                dst.from_proto(bp, reshape=False)

    def to_proto(self, *, write_blobs: bool = True) -> caffe_pb.NetParameter:
        out = caffe_pb.NetParameter()
        out.CopyFrom(self.param)
        if write_blobs:
            for lp, layer in zip(out.layer, self.layers):
                lp.blobs = [b.to_proto() for b in layer.blobs]
        return out


def net_from_prototxt(path: str, **kwargs) -> Net:
    return Net(text_format.parse_file(path, caffe_pb.NetParameter), **kwargs)
