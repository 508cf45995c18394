"""Vision layer catalog: Convolution, Pooling, LRN, InnerProduct, activations,
Dropout, Softmax(+Loss), Accuracy, BatchNorm/Scale/Bias.

Semantics match upstream Caffe (reference layer census: SURVEY.md §2.5 row
"Layer catalog", §3.6 kernel table).  All math dispatches through
`caffeonspark_amd.ops` so the GPU path runs gfx950 HIP kernels.
"""

from __future__ import annotations

import torch

from ... import ops
from ...proto import caffe_pb
from ..blob import Blob
from .base import Layer, register_layer


_REP_NAME = {"kernel": "kernel_size", "stride": "stride", "pad": "pad"}


def _resolve_hw(p, field: str, default=0):
    """Resolve kernel/stride/pad from the repeated (or scalar) field or the
    explicit _h/_w pair."""
    h = getattr(p, field + "_h")
    w = getattr(p, field + "_w")
    if h or w:
        return int(h), int(w)
    rep = getattr(p, _REP_NAME[field])
    if isinstance(rep, (int, float)):
        if rep:
            return int(rep), int(rep)
    elif rep:
        if len(rep) == 1:
            return int(rep[0]), int(rep[0])
        return int(rep[0]), int(rep[1])
    return default, default


@register_layer("Convolution")
class ConvolutionLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.convolution_param
        self.num_output = int(p.num_output)
        self.kh, self.kw = _resolve_hw(p, "kernel")
        self.sh, self.sw = _resolve_hw(p, "stride", 1)
        if self.sh == 0:
            self.sh = self.sw = 1
        self.ph, self.pw = _resolve_hw(p, "pad", 0)
        dil = p.dilation
        self.dil = int(dil[0]) if dil else 1
        self.groups = int(p.group)
        self.bias_term = p.bias_term
        cin = bottom[0].shape[1]
        assert cin % self.groups == 0 and self.num_output % self.groups == 0
        self.add_param([self.num_output, cin // self.groups, self.kh, self.kw],
                       p.weight_filler, name=self.name + "_w")
        if self.bias_term:
            self.add_param([self.num_output], p.bias_filler,
                           name=self.name + "_b")

    def forward(self, bottom, top):
        x = bottom[0].data
        w = self.weight(0)
        b = self.blobs[1].data if self.bias_term else None  # fp32 master
        self._ctx = {}
        kw = {}
        co = getattr(self, "_concat_out", None)
        if co is not None and x.is_cuda and x.dtype == torch.bfloat16:
            concat_l, c_off = co
            eff_h = (self.kh - 1) * self.dil + 1
            eff_w = (self.kw - 1) * self.dil + 1
            P = (x.shape[2] + 2 * self.ph - eff_h) // self.sh + 1
            Q = (x.shape[3] + 2 * self.pw - eff_w) // self.sw + 1
            kw["out_into"] = (concat_l.ensure_buf(
                x.shape[0], P, Q, torch.bfloat16, x.device), c_off)
        top[0].data = ops.conv2d_forward(
            x, w, b, (self.sh, self.sw), (self.ph, self.pw),
            (self.dil, self.dil), self.groups, ctx=self._ctx,
            relu=getattr(self, "_fuse_relu", False), **kw)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        dy = top[0].diff
        x = bottom[0].data
        w = self.weight(0)
        need_dw = self.blobs[0]._lr_mult != 0
        need_db = self.bias_term and self.blobs[1]._lr_mult != 0
        if self.__dict__.pop("_db_done_step", False):
            need_db = False      # fused into the ReLU backward pass
        wb = self.blobs[0]
        dw_out = None
        if need_dw and getattr(wb, "_grad_virgin", False):
            d = wb.ensure_diff()
            if d.dtype == torch.float32 and d.is_contiguous():
                dw_out = d.view(wb.shape)
        db_out = self._bias_arena() if need_db else None
        # branch fan-in (inception): when the bottom already has a diff,
        # let the dx GEMM epilogue accumulate into it in place
        dxb = bottom[0].diff
        dx_into = dxb if (propagate_down[0] and dxb is not None
                          and x.is_cuda and dxb.dtype == torch.bfloat16
                          and list(dxb.shape) == list(x.shape)) else None
        dx, dw, db = ops.conv2d_backward(
            x, w, dy, (self.sh, self.sw), (self.ph, self.pw),
            (self.dil, self.dil), self.groups,
            need_dx=propagate_down[0], need_dw=need_dw, bias=need_db,
            ctx=getattr(self, "_ctx", None), dw_out=dw_out, db_out=db_out,
            dx_into=dx_into)
        if dw is not None:
            if dw is dw_out:
                wb._grad_virgin = False
            else:
                self.acc_param_diff(0, dw)
        if db is not None and db is not db_out:
            self.acc_param_diff(1, db)
        if propagate_down[0] and dx is not dx_into:
            self.acc_blob_diff(bottom[0], dx, False)


@register_layer("InnerProduct")
class InnerProductLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.inner_product_param
        self.num_output = int(p.num_output)
        self.axis = p.axis
        self.bias_term = p.bias_term
        self.transpose = p.transpose
        k = 1
        for d in bottom[0].shape[self.axis:]:
            k *= d
        self.k = k
        wshape = [k, self.num_output] if self.transpose else [self.num_output, k]
        self.add_param(wshape, p.weight_filler, name=self.name + "_w")
        if self.bias_term:
            self.add_param([self.num_output], p.bias_filler,
                           name=self.name + "_b")

    def _flatten(self, t):
        m = 1
        for d in t.shape[:self.axis]:
            m *= d
        return t.reshape(m, self.k)

    def forward(self, bottom, top):
        x = self._flatten(bottom[0].data)
        w = self.weight(0)
        if self.transpose:
            w = w.t()
        b = self.blobs[1].data if self.bias_term else None  # fp32 master
        y = ops.fc_forward(x, w, b, relu=getattr(self, "_fuse_relu", False))
        out_shape = list(bottom[0].shape[:self.axis]) + [self.num_output]
        top[0].data = y.reshape(out_shape)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        dy = top[0].diff.reshape(-1, self.num_output)
        x = self._flatten(bottom[0].data)
        w = self.weight(0)
        if self.transpose:
            w = w.t()
        need_dw = self.blobs[0]._lr_mult != 0
        wb = self.blobs[0]
        dw_out = None
        if need_dw and not self.transpose and \
                getattr(wb, "_grad_virgin", False):
            # first gradient write this step: let the dw GEMM store
            # straight into the fp32 arena slice (no dwp->arena copy)
            d = wb.ensure_diff()
            if d.dtype == torch.float32 and d.is_contiguous():
                dw_out = d.view(wb.shape)
        need_db = self.bias_term and self.blobs[1]._lr_mult != 0
        if self.__dict__.pop("_db_done_step", False):
            need_db = False      # fused into the ReLU backward pass
        db_out = self._bias_arena() if need_db else None
        dx, dw, db = ops.fc_backward(x, w, dy,
                                     need_dx=propagate_down[0],
                                     bias=need_db, dw_out=dw_out,
                                     db_out=db_out)
        if dw is not None and need_dw:
            if dw is dw_out:
                wb._grad_virgin = False
            else:
                self.acc_param_diff(0, dw.t() if self.transpose else dw)
        if db is not None and db is not db_out:
            self.acc_param_diff(1, db)
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0], dx.reshape(bottom[0].data.shape), False)


@register_layer("ReLU")
class ReLULayer(Layer):
    def setup(self, bottom, top):
        self.slope = self.param.relu_param.negative_slope

    def forward(self, bottom, top):
        if getattr(self, "_fused_upstream", False):
            top[0].data = bottom[0].data  # producer already applied ReLU
            return 0.0
        top[0].data = ops.relu_forward(bottom[0].data, self.slope)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if not propagate_down[0]:
            return
        if self.__dict__.pop("_mask_done_step", False):
            # the fused concat already applied this ReLU's mask densely
            self.acc_blob_diff(bottom[0], top[0].diff,
                               top[0] is bottom[0])
            return
        prod = getattr(self, "_db_producer", None)
        db_out = None
        import os as _os
        if prod is not None and self.net._ops_mode == "bf16" \
                and bool(int(_os.environ.get("COS_RELU_DB", "0"))):
            # fused relu'+colsum measured SLOWER than the two separate
            # passes (AlexNet 54.1k->52.6k, GoogLeNet 10.77k->10.30k,
            # same-box A/B): the column-strip block shape the reduction
            # needs costs more dx-write coalescing than the saved dy
            # read.  Kept behind COS_RELU_DB=1 for evidence.
            db_out = prod._bias_arena()
        if db_out is not None:
            dx, done = ops.relu_backward(top[0].data, top[0].diff,
                                         self.slope, db_out=db_out)
            if done:
                prod._db_done_step = True
        else:
            dx = ops.relu_backward(top[0].data, top[0].diff, self.slope)
        self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])


@register_layer("Sigmoid")
class SigmoidLayer(Layer):
    def forward(self, bottom, top):
        top[0].data = ops.sigmoid_forward(bottom[0].data)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = ops.sigmoid_backward(top[0].data, top[0].diff)
            self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])


@register_layer("TanH")
class TanHLayer(Layer):
    def forward(self, bottom, top):
        top[0].data = ops.tanh_forward(bottom[0].data)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = ops.tanh_backward(top[0].data, top[0].diff)
            self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])


@register_layer("Pooling")
class PoolingLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.pooling_param
        self.method = p.pool
        self.global_pooling = p.global_pooling
        if self.global_pooling:
            self.kh = self.kw = 0
        else:
            self.kh, self.kw = _resolve_hw(p, "kernel")
        self.sh, self.sw = _resolve_hw(p, "stride", int(p.stride))
        if self.sh == 0:
            self.sh = self.sw = 1
        self.ph, self.pw = _resolve_hw(p, "pad", int(p.pad))
        self._idx = None

    def forward(self, bottom, top):
        x = bottom[0].data
        if self.global_pooling:
            if self.method == caffe_pb.PoolingParameter.PoolMethod.AVE:
                top[0].data = ops.global_avgpool_forward(x)
            else:
                top[0].data, self._idx = ops.maxpool_forward(
                    x, (x.shape[2], x.shape[3]), (1, 1), (0, 0))
            return 0.0
        k, s, p = (self.kh, self.kw), (self.sh, self.sw), (self.ph, self.pw)
        if self.method == caffe_pb.PoolingParameter.PoolMethod.AVE:
            top[0].data = ops.avgpool_forward(x, k, s, p)
        elif self.method == caffe_pb.PoolingParameter.PoolMethod.MAX:
            top[0].data, self._idx = ops.maxpool_forward(x, k, s, p)
        else:
            raise NotImplementedError("STOCHASTIC pooling")
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if not propagate_down[0]:
            return
        x = bottom[0].data
        dy = top[0].diff
        if self.global_pooling and \
                self.method == caffe_pb.PoolingParameter.PoolMethod.AVE:
            dx = ops.global_avgpool_backward(list(x.shape), dy)
        elif self.method == caffe_pb.PoolingParameter.PoolMethod.AVE:
            dx = ops.avgpool_backward(x, (self.kh, self.kw),
                                      (self.sh, self.sw),
                                      (self.ph, self.pw), dy)
        else:
            dx = ops.maxpool_backward(list(x.shape), self._idx, dy)
        self.acc_blob_diff(bottom[0], dx, False)


@register_layer("LRN")
class LRNLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.lrn_param
        if p.norm_region != caffe_pb.LRNParameter.NormRegion.ACROSS_CHANNELS:
            raise NotImplementedError("WITHIN_CHANNEL LRN")
        self.local_size = int(p.local_size)
        self.alpha = p.alpha
        self.beta = p.beta
        self.k = p.k
        self._scale = None

    def forward(self, bottom, top):
        y, self._scale = ops.lrn_forward(bottom[0].data, self.local_size,
                                         self.alpha, self.beta, self.k)
        top[0].data = y
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = ops.lrn_backward(bottom[0].data, top[0].data, self._scale,
                                  top[0].diff, self.local_size, self.alpha,
                                  self.beta)
            self.acc_blob_diff(bottom[0], dx, False)


@register_layer("Dropout")
class DropoutLayer(Layer):
    def setup(self, bottom, top):
        self.ratio = self.param.dropout_param.dropout_ratio
        self._mask = None

    def forward(self, bottom, top):
        if self.phase == caffe_pb.Phase.TRAIN:
            top[0].data, self._mask = ops.dropout_forward(
                bottom[0].data, self.ratio, generator=self.net.generator)
        else:
            top[0].data = bottom[0].data
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            if self.phase == caffe_pb.Phase.TRAIN and self._mask is not None:
                dx = ops.dropout_backward(self._mask, top[0].diff)
            else:
                dx = top[0].diff
            self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])


@register_layer("Softmax")
class SoftmaxLayer(Layer):
    def setup(self, bottom, top):
        self.axis = self.param.softmax_param.axis

    def forward(self, bottom, top):
        top[0].data = ops.softmax_forward(bottom[0].data, self.axis)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            dx = ops.softmax_backward(top[0].data, top[0].diff, self.axis)
            self.acc_blob_diff(bottom[0], dx, top[0] is bottom[0])


@register_layer("SoftmaxWithLoss")
class SoftmaxWithLossLayer(Layer):
    def setup(self, bottom, top):
        lp = self.param.loss_param
        self.ignore_label = lp.ignore_label if lp.has_field("ignore_label") else None
        self.normalization = lp.normalization
        if lp.has_field("normalize") and not lp.normalize:
            self.normalization = caffe_pb.LossParameter.Normalization.BATCH_SIZE
        self.axis = self.param.softmax_param.axis
        self._prob = None
        self._count = 0

    def _norm_denom(self, x):
        N = caffe_pb.LossParameter.Normalization
        outer = x.shape[0]
        full = x.numel() // x.shape[self.axis]
        if self.normalization == N.FULL:
            return full
        if self.normalization == N.BATCH_SIZE:
            return outer
        if self.normalization == N.NONE:
            return 1
        return max(1, self._count)  # VALID

    def forward(self, bottom, top):
        x, label = bottom[0].data, bottom[1].data
        loss_sum, self._prob, self._count = ops.softmax_loss_forward(
            x, label, self.ignore_label, self.axis)
        loss = loss_sum / self._norm_denom(x)
        top[0].data = loss.detach().reshape(())
        return 0.0  # loss contribution handled by Net via loss_weight

    def backward(self, top, propagate_down, bottom):
        if len(propagate_down) > 1 and propagate_down[1]:
            raise RuntimeError("SoftmaxWithLoss cannot backprop to labels")
        if propagate_down[0]:
            lw = getattr(top[0], "_loss_weight", None)
            if lw is None:
                lw = float(top[0].diff.reshape(-1)[0]) \
                    if top[0].diff is not None else 1.0
            scale = lw / self._norm_denom(bottom[0].data)
            dx = ops.softmax_loss_backward(self._prob, bottom[1].data,
                                           self.ignore_label, scale, self.axis)
            self.acc_blob_diff(bottom[0], dx.to(bottom[0].data.dtype), False)


@register_layer("Accuracy")
class AccuracyLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.accuracy_param
        self.top_k = int(p.top_k)
        self.ignore_label = p.ignore_label if p.has_field("ignore_label") else None

    def forward(self, bottom, top):
        hits, count = ops.accuracy(bottom[0].data, bottom[1].data,
                                   self.top_k, self.ignore_label,
                                   axis=self.param.accuracy_param.axis)
        top[0].data = (hits / max(1, count)).reshape(())
        return 0.0

    def backward(self, top, propagate_down, bottom):
        pass  # no gradients


@register_layer("BatchNorm")
class BatchNormLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.batch_norm_param
        self.eps = p.eps
        self.maf = p.moving_average_fraction
        self.use_global = (p.use_global_stats if p.has_field("use_global_stats")
                           else self.phase == caffe_pb.Phase.TEST)
        c = bottom[0].shape[1]
        # caffe batchnorm blobs: mean, variance, scale_factor (all lr_mult 0)
        self.add_param([c], name=self.name + "_mean")
        self.add_param([c], name=self.name + "_var")
        self.add_param([1], name=self.name + "_sf")
        for b in self.blobs:
            b._lr_mult = 0.0
            b._decay_mult = 0.0
        self._cache = None

    def forward(self, bottom, top):
        x = bottom[0].data
        if self.use_global:
            sf = float(self.blobs[2].data.float())
            scale = 0 if sf == 0 else 1.0 / sf
            mean = (self.blobs[0].data.float() * scale).contiguous()
            var = (self.blobs[1].data.float() * scale).contiguous()
            y, inv_std = ops.bn_forward_infer(x, mean, var, self.eps)
        else:
            y, mean, var, inv_std = ops.bn_forward_train(x, self.eps)
            m = x.numel() / x.shape[1]
            self.blobs[0].data.mul_(self.maf).add_(
                mean.to(self.blobs[0].data.dtype))
            bias_corr = m / max(1.0, m - 1.0)
            self.blobs[1].data.mul_(self.maf).add_(
                (var * bias_corr).to(self.blobs[1].data.dtype))
            self.blobs[2].data.mul_(self.maf).add_(1.0)
        # caffe BN has no affine term, so the output IS xhat — but a
        # downstream in-place layer (the standard BN->Scale->ReLU idiom)
        # replaces the shared blob's data before our backward runs, so
        # hold a reference to the xhat tensor itself (Caffe's x_norm_;
        # layers here assign fresh tensors, never mutate, so the held
        # reference stays valid)
        self._cache = (inv_std, y)
        top[0].data = y
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if not propagate_down[0]:
            return
        inv_std, xhat = self._cache
        dx = ops.bn_backward(xhat, top[0].diff, inv_std,
                             train=not self.use_global)
        self.acc_blob_diff(bottom[0], dx.to(bottom[0].data.dtype),
                           top[0] is bottom[0])


@register_layer("Scale")
class ScaleLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.scale_param
        self.axis = p.axis
        self.bias_term = p.bias_term
        if len(bottom) == 1:
            c = bottom[0].shape[self.axis]
            filler = p.filler if p.has_field("filler") else \
                caffe_pb.FillerParameter(type="constant", value=1.0)
            self.add_param([c], filler, name=self.name + "_scale")
        if self.bias_term:
            self.add_param([bottom[0].shape[self.axis]], p.bias_filler,
                           name=self.name + "_bias")

    def _shape(self, x):
        return [1] * self.axis + [-1] + [1] * (x.dim() - self.axis - 1)

    def forward(self, bottom, top):
        x = bottom[0].data
        # stash the input tensor: when this layer runs in-place
        # (top is bottom — the BN->Scale idiom), assigning top[0].data
        # below repoints the shared blob at y, and backward's dscale
        # needs the original x (Caffe ScaleLayer stashes its in-place
        # bottom the same way)
        self._x = x
        scale = bottom[1].data if len(bottom) > 1 else self.weight(0)
        y = x * scale.reshape(self._shape(x))
        if self.bias_term:
            y = y + self.cast(self.blobs[-1].data).reshape(self._shape(x))
        top[0].data = y
        return 0.0

    def backward(self, top, propagate_down, bottom):
        x = self._x
        dy = top[0].diff
        dims = [d for d in range(x.dim()) if d != self.axis]
        if len(bottom) == 1:
            self.acc_param_diff(0, (dy * x).sum(dim=dims))
            scale = self.weight(0)
        else:
            if propagate_down[1]:
                self.acc_blob_diff(bottom[1], (dy * x).sum(dim=dims), False)
            scale = bottom[1].data
        if self.bias_term:
            self.acc_param_diff(len(self.blobs) - 1, dy.sum(dim=dims))
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0], dy * scale.reshape(self._shape(x)),
                               top[0] is bottom[0])


@register_layer("Bias")
class BiasLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.bias_param
        self.axis = p.axis
        if len(bottom) == 1:
            c = bottom[0].shape[self.axis]
            self.add_param([c], p.filler, name=self.name + "_bias")

    def forward(self, bottom, top):
        x = bottom[0].data
        b = bottom[1].data if len(bottom) > 1 else self.weight(0)
        top[0].data = ops.bias_add(x, b, self.axis)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        dy = top[0].diff
        dims = [d for d in range(dy.dim()) if d != self.axis]
        if len(bottom) == 1:
            self.acc_param_diff(0, dy.sum(dim=dims))
        elif propagate_down[1]:
            self.acc_blob_diff(bottom[1], dy.sum(dim=dims), False)
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0], dy, top[0] is bottom[0])
