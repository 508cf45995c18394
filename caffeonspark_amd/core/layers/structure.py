"""Structural layers: Concat, Slice, Eltwise, Flatten, Reshape, Split,
Silence, Power, Exp, Log — plumbing needed by GoogLeNet (Concat/Inception)
and LRCN (Silence) per SURVEY.md §2.5/§3.6.
"""

from __future__ import annotations

import torch

from ...proto import caffe_pb
from .base import Layer, register_layer


@register_layer("Concat")
class ConcatLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.concat_param
        self.axis = p.axis if p.has_field("axis") or not p.has_field("concat_dim") \
            else int(p.concat_dim)

    def ensure_buf(self, N, P, Q, dtype, device):
        """Fused mode: the shared output buffer the branch convs write
        their channel windows into (allocated by the first producer each
        step, consumed by forward())."""
        buf = getattr(self, "_fused_buf", None)
        if buf is None or buf.shape[0] != N or buf.shape[2] != P \
                or buf.shape[3] != Q:
            buf = torch.empty((N, self._fused_ctot, P, Q), dtype=dtype,
                              device=device,
                              memory_format=torch.channels_last)
            self._fused_buf = buf
        return buf

    def forward(self, bottom, top):
        if getattr(self, "_fused", False) and self._fused_buf is not None:
            # producers already wrote their slices: adopt, no copy
            top[0].data = self._fused_buf
            self._fused_buf = None      # next step reallocates
            return 0.0
        top[0].data = torch.cat([b.data for b in bottom], dim=self.axis)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if getattr(self, "_fused", False):
            relus = getattr(self, "_branch_relus", None)
            dy = top[0].diff
            if relus is not None and self.net._ops_mode == "bf16" \
                    and dy is not None and dy.dtype == torch.bfloat16:
                # every branch ends in a fused ReLU: ONE dense masked
                # pass over the whole concat (full cache lines), then
                # hand out pre-masked windows; the branch ReLU layers
                # pass straight through
                from ... import ops
                dxm = ops.relu_backward(top[0].data, dy)
                off = 0
                for b, pd, rl in zip(bottom, propagate_down, relus):
                    k = b.data.shape[1]
                    if pd:
                        self.acc_blob_diff(b, dxm[:, off:off + k], False)
                        rl._mask_done_step = True
                    off += k
                return
            # hand each branch a channel-slice VIEW — the in-place ReLU
            # backward consumes it stride-aware, so no slicing copy
            off = 0
            for b, pd in zip(bottom, propagate_down):
                k = b.data.shape[1]
                if pd:
                    self.acc_blob_diff(b, dy[:, off:off + k], False)
                off += k
            return
        sizes = [b.data.shape[self.axis] for b in bottom]
        pieces = torch.split(top[0].diff, sizes, dim=self.axis)
        for b, pd, piece in zip(bottom, propagate_down, pieces):
            if pd:
                if piece.dim() == 4:
                    piece = piece.contiguous(
                        memory_format=torch.channels_last)
                else:
                    piece = piece.contiguous()
                self.acc_blob_diff(b, piece, False)


@register_layer("Slice")
class SliceLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.slice_param
        self.axis = p.axis
        self.points = [int(x) for x in p.slice_point]

    def _sizes(self, x):
        total = x.shape[self.axis]
        if self.points:
            pts = [0] + self.points + [total]
            return [pts[i + 1] - pts[i] for i in range(len(pts) - 1)]
        n = len(self.param.top)
        assert total % n == 0
        return [total // n] * n

    def forward(self, bottom, top):
        pieces = torch.split(bottom[0].data, self._sizes(bottom[0].data),
                             dim=self.axis)
        for t, piece in zip(top, pieces):
            if piece.dim() == 4:
                t.data = piece.contiguous(memory_format=torch.channels_last)
            else:
                t.data = piece.contiguous()
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = torch.cat([t.diff if t.diff is not None
                            else torch.zeros_like(t.data) for t in top],
                           dim=self.axis)
            self.acc_blob_diff(bottom[0], dx, False)


@register_layer("Eltwise")
class EltwiseLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.eltwise_param
        self.op = p.operation
        self.coeff = list(p.coeff) or [1.0] * len(self.param.bottom)
        self._argmax = None

    def forward(self, bottom, top):
        Op = caffe_pb.EltwiseParameter.EltwiseOp
        if self.op == Op.SUM:
            y = bottom[0].data * self.coeff[0]
            for b, c in zip(bottom[1:], self.coeff[1:]):
                y = y + b.data * c
        elif self.op == Op.PROD:
            y = bottom[0].data
            for b in bottom[1:]:
                y = y * b.data
        else:  # MAX
            stacked = torch.stack([b.data for b in bottom])
            y, self._argmax = stacked.max(dim=0)
        top[0].data = y
        return 0.0

    def backward(self, top, propagate_down, bottom):
        Op = caffe_pb.EltwiseParameter.EltwiseOp
        dy = top[0].diff
        for i, (b, pd) in enumerate(zip(bottom, propagate_down)):
            if not pd:
                continue
            if self.op == Op.SUM:
                dx = dy * self.coeff[i]
            elif self.op == Op.PROD:
                dx = dy.clone()
                for j, ob in enumerate(bottom):
                    if j != i:
                        dx = dx * ob.data
            else:
                dx = dy * (self._argmax == i).to(dy.dtype)
            self.acc_blob_diff(b, dx, False)


@register_layer("Flatten")
class FlattenLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.flatten_param
        self.axis = p.axis
        self.end_axis = p.end_axis

    def forward(self, bottom, top):
        x = bottom[0].data
        end = self.end_axis if self.end_axis >= 0 else x.dim() + self.end_axis
        shape = list(x.shape[:self.axis]) + [-1] + list(x.shape[end + 1:])
        top[0].data = x.reshape(shape)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0],
                               top[0].diff.reshape(bottom[0].data.shape), False)


@register_layer("Reshape")
class ReshapeLayer(Layer):
    def setup(self, bottom, top):
        self.dims = [int(d) for d in self.param.reshape_param.shape.dim]

    def forward(self, bottom, top):
        x = bottom[0].data
        shape = []
        for i, d in enumerate(self.dims):
            if d == 0:
                shape.append(x.shape[i])
            else:
                shape.append(d)
        top[0].data = x.reshape(shape)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0],
                               top[0].diff.reshape(bottom[0].data.shape), False)


@register_layer("Split")
class SplitLayer(Layer):
    """Explicit fan-out (Caffe inserts these automatically; our Net
    accumulates diffs instead, but the layer type is still supported)."""

    def forward(self, bottom, top):
        for t in top:
            t.data = bottom[0].data
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = None
            for t in top:
                if t.diff is not None:
                    dx = t.diff.clone() if dx is None else dx + t.diff
            if dx is not None:
                self.acc_blob_diff(bottom[0], dx, False)


@register_layer("Silence")
class SilenceLayer(Layer):
    def forward(self, bottom, top):
        return 0.0

    def backward(self, top, propagate_down, bottom):
        for b, pd in zip(bottom, propagate_down):
            if pd:
                b.ensure_diff()  # zeros


@register_layer("Power")
class PowerLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.power_param
        self.power, self.scale, self.shift = p.power, p.scale, p.shift

    def forward(self, bottom, top):
        x = bottom[0].data
        inner = self.scale * x + self.shift
        top[0].data = inner if self.power == 1.0 else inner.pow(self.power)
        self._inner = inner
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dy = top[0].diff
            if self.power == 1.0:
                dx = dy * self.scale
            else:
                dx = dy * self.power * self.scale * self._inner.pow(self.power - 1)
            self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])
