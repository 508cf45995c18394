"""Recurrent layers: LSTM (time-major), Embed.

Caffe's LSTM layer (used by the reference's LRCN captioning net,
data/lrcn_cos.prototxt — SURVEY.md §5 "Long-context") consumes
time-major input x:[T,N,D] plus continuation markers cont:[T,N] and
produces h:[T,N,H].  Upstream implements it by unrolling an internal net;
here it is implemented directly: one big input GEMM over all timesteps,
then a per-step recurrent GEMM + fused LSTM-unit op (the HIP kernel target
for the recurrent path).

Params (upstream-compatible blob order):
  0: W_xc [4H, D]   input-to-gate weights (gate order i,f,o,g)
  1: b_c  [4H]      gate bias
  2: W_hc [4H, H]   hidden-to-gate weights
"""

from __future__ import annotations

import torch

from ... import ops
from .base import Layer, register_layer


@register_layer("LSTM")
class LSTMLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.recurrent_param
        self.h = int(p.num_output)
        if p.expose_hidden:
            raise NotImplementedError("LSTM expose_hidden")
        x_shape = bottom[0].shape
        self.t_axis, self.n_axis = 0, 1
        d = 1
        for dim in x_shape[2:]:
            d *= dim
        self.d = d
        self.static = len(bottom) > 2  # x_static bottom
        self.add_param([4 * self.h, d], p.weight_filler, name=self.name + "_Wxc")
        self.add_param([4 * self.h], p.bias_filler, name=self.name + "_bc")
        self.add_param([4 * self.h, self.h], p.weight_filler,
                       name=self.name + "_Whc")
        if self.static:
            self.add_param([4 * self.h, self.d_static(bottom)],
                           p.weight_filler, name=self.name + "_Wxsc")
        self._cache = None

    def d_static(self, bottom):
        ds = 1
        for dim in bottom[2].shape[1:]:
            ds *= dim
        return ds

    def forward(self, bottom, top):
        x = bottom[0].data
        cont = bottom[1].data
        T, N = x.shape[0], x.shape[1]
        H = self.h
        w_xc, b_c, w_hc = (self.weight(0), self.weight(1), self.weight(2))
        xg = ops.fc_forward(x.reshape(T * N, -1), w_xc, b_c).reshape(T, N, 4 * H)
        if self.static:
            xs = ops.fc_forward(bottom[2].data.reshape(N, -1), self.weight(3),
                                None)
            xg = xg + xs.unsqueeze(0)
        seq_fwd = ops.gpu_op("lstm_seq_forward") if x.is_cuda else None
        if seq_fwd is not None:
            # whole recurrence driven from C++ (3 kernel launches/step)
            y, seq_cache = seq_fwd(xg, w_hc, cont)
            top[0].data = y
            self._cache = ("seq", x, cont, seq_cache)
            return 0.0
        h_prev = torch.zeros(N, H, dtype=x.dtype, device=x.device)
        c_prev = torch.zeros(N, H, dtype=torch.float32, device=x.device)
        hs, caches, h_prevs, c_prevs = [], [], [], []
        for t in range(T):
            cont_t = cont[t].reshape(N, 1).to(x.dtype)
            h_in = h_prev * cont_t
            gates = xg[t] + ops.fc_forward(h_in, w_hc, None)
            c_t, h_t, cache = ops.lstm_unit_forward(c_prev, gates, cont[t])
            h_prevs.append(h_in)
            c_prevs.append(c_prev)
            caches.append(cache)
            hs.append(h_t)
            c_prev = c_t.float()
            h_prev = h_t
        y = torch.stack(hs, dim=0)
        top[0].data = y
        self._cache = (x, cont, caches, h_prevs, c_prevs, hs)
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if self._cache and isinstance(self._cache[0], str) \
                and self._cache[0] == "seq":
            self._backward_seq(top, propagate_down, bottom)
            return
        x, cont, caches, h_prevs, c_prevs, hs = self._cache
        T, N = x.shape[0], x.shape[1]
        H = self.h
        w_xc, w_hc = self.weight(0), self.weight(2)
        dy = top[0].diff
        d_xg = torch.zeros(T, N, 4 * H, dtype=torch.float32, device=x.device)
        dw_hc = torch.zeros_like(self.blobs[2].data)
        dh_next = torch.zeros(N, H, dtype=torch.float32, device=x.device)
        dc_next = torch.zeros(N, H, dtype=torch.float32, device=x.device)
        for t in reversed(range(T)):
            dh = (dy[t].float() if dy is not None else 0) + dh_next
            dc_prev, d_gates = ops.lstm_unit_backward(
                c_prevs[t], caches[t], dc_next, dh.to(x.dtype))
            d_gates_f = d_gates.float()
            d_xg[t] = d_gates_f
            # gates_t = xg[t] + h_in @ w_hc^T ; h_in = h_{t-1} * cont_t
            dh_in = d_gates_f @ w_hc.float()
            dw_hc += d_gates_f.t() @ h_prevs[t].float()
            cont_t = cont[t].reshape(N, 1).float()
            dh_next = dh_in * cont_t
            dc_next = dc_prev
        # input GEMM backward
        d_xg_flat = d_xg.reshape(T * N, 4 * H)
        if propagate_down[0]:
            dx = (d_xg_flat @ w_xc.float()).reshape(x.shape)
            self.acc_blob_diff(bottom[0], dx.to(x.dtype), False)
        if self.blobs[0]._lr_mult != 0:
            self.acc_param_diff(0, d_xg_flat.t() @ x.reshape(T * N, -1).float())
            self.acc_param_diff(1, d_xg_flat.sum(0))
            self.acc_param_diff(2, dw_hc)
        if self.static and self.blobs[3]._lr_mult != 0:
            xs = bottom[2].data.reshape(N, -1).float()
            d_static = d_xg.sum(dim=0)  # [N, 4H]
            self.acc_param_diff(3, d_static.t() @ xs)
            if propagate_down[2]:
                dxs = d_static @ self.weight(3).float()
                self.acc_blob_diff(bottom[2],
                                   dxs.reshape(bottom[2].data.shape).to(x.dtype),
                                   False)


    def _backward_seq(self, top, propagate_down, bottom):
        _, x, cont, seq_cache = self._cache
        T, N = x.shape[0], x.shape[1]
        H = self.h
        dy = top[0].diff
        seq_bwd = ops.gpu_op("lstm_seq_backward")
        dxg, dw_hc = seq_bwd(dy.reshape(T, N, H), self.weight(2), seq_cache)
        dxg_flat = dxg.reshape(T * N, 4 * H)
        x_flat = x.reshape(T * N, -1)
        # shared tail: input GEMM grads == a linear layer's backward
        dx, dw_xc, db = ops.fc_backward(
            x_flat, self.weight(0), dxg_flat,
            need_dx=propagate_down[0], bias=True)
        if self.blobs[0]._lr_mult != 0:
            self.acc_param_diff(0, dw_xc)
            self.acc_param_diff(1, db)
            self.acc_param_diff(2, dw_hc)
        if propagate_down[0]:
            self.acc_blob_diff(bottom[0], dx.reshape(x.shape), False)
        if self.static and self.blobs[3]._lr_mult != 0:
            xs = bottom[2].data.reshape(N, -1)
            d_static = dxg.float().sum(dim=0)  # [N, 4H]
            self.acc_param_diff(3, d_static.t() @ xs.float())
            if propagate_down[2]:
                dxs = (d_static @ self.weight(3).float()).to(x.dtype)
                self.acc_blob_diff(bottom[2],
                                   dxs.reshape(bottom[2].data.shape), False)


@register_layer("Embed")
class EmbedLayer(Layer):
    def setup(self, bottom, top):
        p = self.param.embed_param
        self.e = int(p.num_output)
        self.v = int(p.input_dim)
        self.bias_term = p.bias_term
        self.add_param([self.v, self.e], p.weight_filler, name=self.name + "_w")
        if self.bias_term:
            self.add_param([self.e], p.bias_filler, name=self.name + "_b")

    def forward(self, bottom, top):
        idx = bottom[0].data
        w = self.weight(0)
        b = self.cast(self.blobs[1].data) if self.bias_term else None
        y = ops.embed_forward(idx.reshape(idx.shape[0], -1) if idx.dim() > 2
                              else idx, w, b)
        # caffe embed output: bottom shape (with trailing singleton dims
        # dropped) + [E]
        base = [s for s in idx.shape]
        # drop trailing singleton SPATIAL dims but never the batch axis
        # (time-major [T, 1] single-image decode must stay 2-d)
        while len(base) > 2 and base[-1] == 1:
            base = base[:-1]
        top[0].data = y.reshape(base + [self.e])
        return 0.0

    def backward(self, top, propagate_down, bottom):
        if propagate_down[0]:
            raise RuntimeError("Embed cannot backprop to indices")
        if self.blobs[0]._lr_mult == 0:
            return
        idx = bottom[0].data
        dy = top[0].diff.reshape(-1, self.e)
        dw, db = ops.embed_backward(idx, dy, self.v, self.bias_term)
        self.acc_param_diff(0, dw)
        if db is not None:
            self.acc_param_diff(1, db)
