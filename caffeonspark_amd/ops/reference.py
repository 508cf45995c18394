"""Pure-torch reference implementations of every compute op.

These are the CPU execution path and the numerics oracle the HIP kernels are
tested against (tests compare the gfx950 kernels to these in fp32).  Layer
semantics follow upstream Caffe (the engine the reference drives through
`caffe::Net`/`caffe::Solver`).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F


# ----------------------------------------------------------------- convolution

def conv2d_forward(x, w, b, stride, pad, dilation, groups, ctx=None,
                   relu=False):
    if b is not None and b.dtype != x.dtype:
        b = b.to(x.dtype)   # callers pass the fp32 master bias
    y = F.conv2d(x, w, b, stride=stride, padding=pad,
                 dilation=dilation, groups=groups)
    return F.relu(y) if relu else y


def conv2d_backward(x, w, dy, stride, pad, dilation, groups,
                    need_dx=True, need_dw=True, bias=True, ctx=None,
                    dw_out=None, db_out=None, dx_into=None):
    # db_out/dx_into: GPU-path hints, unused here
    dx = dw = db = None
    if need_dx:
        dx = torch.nn.grad.conv2d_input(list(x.shape), w, dy, stride=stride,
                                        padding=pad, dilation=dilation,
                                        groups=groups)
    if need_dw:
        dw = torch.nn.grad.conv2d_weight(x, list(w.shape), dy, stride=stride,
                                         padding=pad, dilation=dilation,
                                         groups=groups)
        if dw_out is not None:
            dw_out.copy_(dw)
            dw = dw_out
    if bias:
        db = dy.sum(dim=(0, 2, 3))
    return dx, dw, db


# ------------------------------------------------------------- inner product

def fc_forward(x, w, b, relu=False):
    """x: [M, K], w: [N, K] (caffe layout), b: [N] or None."""
    if b is not None and b.dtype != x.dtype:
        b = b.to(x.dtype)   # callers pass the fp32 master bias
    y = F.linear(x, w, b)
    return F.relu(y) if relu else y


def fc_backward(x, w, dy, need_dx=True, bias=True, dw_out=None,
                db_out=None):  # db_out: GPU-path hint, unused here
    dx = dy @ w if need_dx else None
    dw = dy.t() @ x
    if dw_out is not None:
        dw_out.copy_(dw)
        dw = dw_out
    db = dy.sum(dim=0) if bias else None
    return dx, dw, db


# ------------------------------------------------------------------ activations

def relu_forward(x, negative_slope=0.0):
    return F.leaky_relu(x, negative_slope) if negative_slope else F.relu(x)


def relu_backward(y, dy, negative_slope=0.0, db_out=None):
    if db_out is not None:
        dx = torch.where(y > 0, dy, dy * negative_slope)
        return dx, False
    """Caffe ReLU backward uses bottom data; with in-place layers only top
    data is available — sign(y) equals sign(x) for slope<1 so using y is
    exact for slope >= 0."""
    if negative_slope:
        return torch.where(y > 0, dy, dy * negative_slope)
    return dy * (y > 0).to(dy.dtype)


def sigmoid_forward(x):
    return torch.sigmoid(x)


def sigmoid_backward(y, dy):
    return dy * y * (1.0 - y)


def tanh_forward(x):
    return torch.tanh(x)


def tanh_backward(y, dy):
    return dy * (1.0 - y * y)


# --------------------------------------------------------------------- pooling

def maxpool_forward(x, kernel, stride, pad):
    y, idx = F.max_pool2d(x, kernel_size=kernel, stride=stride, padding=pad,
                          ceil_mode=True, return_indices=True)
    return y, idx


def maxpool_backward(x_shape, idx, dy):
    n, c = dy.shape[0], dy.shape[1]
    dx = torch.zeros(n, c, x_shape[2] * x_shape[3], dtype=dy.dtype,
                     device=dy.device)
    dx.scatter_add_(2, idx.reshape(n, c, -1), dy.reshape(n, c, -1))
    return dx.reshape(x_shape)


def avgpool_forward(x, kernel, stride, pad):
    # caffe divides by the window clipped at the *padded* boundary
    return F.avg_pool2d(x, kernel_size=kernel, stride=stride, padding=pad,
                        ceil_mode=True, count_include_pad=True)


def avgpool_backward(x, kernel, stride, pad, dy):
    x_req = x.detach().requires_grad_(True)
    with torch.enable_grad():
        y = avgpool_forward(x_req, kernel, stride, pad)
    return torch.autograd.grad(y, x_req, dy)[0]


def global_avgpool_forward(x):
    return x.mean(dim=(2, 3), keepdim=True)


def global_avgpool_backward(x_shape, dy):
    scale = 1.0 / (x_shape[2] * x_shape[3])
    return (dy * scale).expand(x_shape[0], x_shape[1], x_shape[2], x_shape[3]).contiguous()


# ------------------------------------------------------------------------- LRN

def lrn_forward(x, local_size, alpha, beta, k):
    """Across-channel LRN (caffe): scale = k + alpha/n * sum_win x^2,
    y = x * scale^-beta. Returns (y, scale) — scale reused in backward."""
    xf = x.float()
    sq = xf * xf
    pad = local_size // 2
    # sum over channel window via avg_pool3d trick
    win = F.avg_pool3d(sq.unsqueeze(1), kernel_size=(local_size, 1, 1),
                       stride=1, padding=(pad, 0, 0),
                       count_include_pad=True).squeeze(1) * local_size
    scale = k + (alpha / local_size) * win
    y = xf * scale.pow(-beta)
    return y.to(x.dtype), scale


def lrn_backward(x, y, scale, dy, local_size, alpha, beta):
    xf, yf, dyf = x.float(), y.float(), dy.float()
    pad = local_size // 2
    ratio = dyf * yf / scale
    win = F.avg_pool3d(ratio.unsqueeze(1), kernel_size=(local_size, 1, 1),
                       stride=1, padding=(pad, 0, 0),
                       count_include_pad=True).squeeze(1) * local_size
    dx = dyf * scale.pow(-beta) - (2.0 * alpha * beta / local_size) * xf * win
    return dx.to(x.dtype)


# --------------------------------------------------------------------- softmax

def softmax_forward(x, axis=1):
    return F.softmax(x.float(), dim=axis).to(x.dtype)


def softmax_backward(y, dy, axis=1):
    yf, dyf = y.float(), dy.float()
    dot = (yf * dyf).sum(dim=axis, keepdim=True)
    return (yf * (dyf - dot)).to(y.dtype)


def softmax_loss_forward(x, label, ignore_label: Optional[int], axis=1):
    """Returns (loss_sum, prob, valid_count). x: [..., C at axis, ...],
    label matches x with the class axis removed."""
    xf = x.float()
    logp = F.log_softmax(xf, dim=axis)
    prob = logp.exp()
    lab = label.long()
    lp = logp.movedim(axis, -1)
    lab_flat = lab.reshape(-1)
    lp_flat = lp.reshape(-1, lp.shape[-1])
    if ignore_label is not None:
        valid = lab_flat != ignore_label
        safe_lab = torch.where(valid, lab_flat, torch.zeros_like(lab_flat))
        picked = lp_flat.gather(1, safe_lab.unsqueeze(1)).squeeze(1)
        loss = -(picked * valid.to(picked.dtype)).sum()
        count = int(valid.sum())
    else:
        picked = lp_flat.gather(1, lab_flat.unsqueeze(1)).squeeze(1)
        loss = -picked.sum()
        count = lab_flat.numel()
    return loss, prob, count


def softmax_loss_backward(prob, label, ignore_label: Optional[int], scale, axis=1):
    lab = label.long()
    # operate with the class axis last, then restore
    p = prob.movedim(axis, -1)
    shape = p.shape
    p2 = p.reshape(-1, shape[-1]).clone()
    lab_flat = lab.reshape(-1, 1)
    if ignore_label is not None:
        valid = lab_flat != ignore_label
        safe = torch.where(valid, lab_flat, torch.zeros_like(lab_flat))
        p2.scatter_add_(1, safe, -torch.ones_like(safe, dtype=p2.dtype))
        p2 = p2 * valid.to(p2.dtype)
    else:
        p2.scatter_add_(1, lab_flat,
                        -torch.ones_like(lab_flat, dtype=p2.dtype))
    return (p2.reshape(shape).movedim(-1, axis) * scale)


# --------------------------------------------------------------------- dropout

def dropout_forward(x, ratio, generator=None):
    keep = 1.0 - ratio
    # draw through the net's seeded generator so solver random_seed makes
    # masks reproducible (upstream Caffe's seeded RNG covers dropout); a
    # CPU generator serving a CUDA tensor (fp32 GPU path) draws host-side
    if generator is not None and generator.device != x.device:
        r = torch.rand(x.shape, dtype=torch.float32,
                       generator=generator).to(x.device)
    else:
        r = torch.rand(x.shape, dtype=torch.float32, device=x.device,
                       generator=generator)
    mask = (r < keep).to(x.dtype) / keep
    return x * mask, mask


def dropout_backward(mask, dy):
    return dy * mask


# ----------------------------------------------------------------------- embed

def embed_forward(idx, w, b=None):
    """idx: int tensor [...], w: [V, E]. Returns [..., E]."""
    out = w[idx.long().clamp_(0, w.shape[0] - 1)]
    if b is not None:
        out = out + b
    return out


def embed_backward(idx, dy, vocab_size, bias=True):
    e = dy.shape[-1]
    dw = torch.zeros(vocab_size, e, dtype=torch.float32, device=dy.device)
    dw.index_add_(0, idx.long().reshape(-1), dy.float().reshape(-1, e))
    db = dy.float().reshape(-1, e).sum(0) if bias else None
    return dw, db


# ------------------------------------------------------------------- LSTM unit

def lstm_unit_forward(c_prev, gates, cont):
    """Caffe LSTMUnit semantics. gates: [N, 4H] pre-activation ordered
    (i, f, o, g); cont: [N] or [N,1] continuation flags; c_prev: [N, H].
    Returns (c, h, cache) — cache holds activated gates for backward."""
    n, h4 = gates.shape
    h = h4 // 4
    g = gates.float().view(n, 4, h)
    i = torch.sigmoid(g[:, 0])
    f = torch.sigmoid(g[:, 1])
    o = torch.sigmoid(g[:, 2])
    gg = torch.tanh(g[:, 3])
    cont_ = cont.float().view(n, 1)
    c = f * c_prev.float() * cont_ + i * gg
    tc = torch.tanh(c)
    hh = o * tc
    return c.to(gates.dtype), hh.to(gates.dtype), (i, f, o, gg, tc, cont_)


def lstm_unit_backward(c_prev, cache, dc_next, dh):
    i, f, o, gg, tc, cont_ = cache
    dhf = dh.float()
    dc = dc_next.float() + dhf * o * (1.0 - tc * tc)
    do = dhf * tc
    di = dc * gg
    dg = dc * i
    df = dc * c_prev.float() * cont_
    dc_prev = dc * f * cont_
    d_gates = torch.stack([
        di * i * (1 - i),
        df * f * (1 - f),
        do * o * (1 - o),
        dg * (1 - gg * gg),
    ], dim=1)  # [N, 4, H]
    n = d_gates.shape[0]
    return dc_prev, d_gates.reshape(n, -1)


# -------------------------------------------------------------------- concat &c

def accuracy(x, label, top_k=1, ignore_label: Optional[int] = None,
             axis: int = -1):
    lab = label.long().reshape(-1)
    if x.dim() > 2 and axis not in (-1, x.dim() - 1):
        x = x.movedim(axis, -1)
    scores = x.reshape(lab.shape[0], -1)
    topk = scores.topk(min(top_k, scores.shape[1]), dim=1).indices
    hit = (topk == lab.unsqueeze(1)).any(dim=1)
    if ignore_label is not None:
        valid = lab != ignore_label
        hit = hit & valid
        count = int(valid.sum())
    else:
        count = lab.shape[0]
    return hit.float().sum(), count


def bias_add(x, b, axis=1):
    shape = [1] * x.dim()
    shape[axis] = -1
    return x + b.reshape(shape)


# ------------------------------------------------------------------ optimizer

def sgd_update(param, grad, momentum_buf, lr, momentum, weight_decay):
    """Caffe SGD: V = mu*V + lr*(grad + wd*param); param -= V (in place)."""
    g = grad
    if weight_decay:
        g = g + weight_decay * param
    momentum_buf.mul_(momentum).add_(g, alpha=lr)
    param.sub_(momentum_buf)


def nesterov_update(param, grad, momentum_buf, lr, momentum, weight_decay):
    g = grad
    if weight_decay:
        g = g + weight_decay * param
    v_prev = momentum_buf.clone()
    momentum_buf.mul_(momentum).add_(g, alpha=lr)
    param.sub_((1 + momentum) * momentum_buf - momentum * v_prev)


def adam_update(param, grad, m, v, lr, beta1, beta2, eps, weight_decay, t):
    g = grad
    if weight_decay:
        g = g + weight_decay * param
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    correction = (1 - beta2 ** t) ** 0.5 / (1 - beta1 ** t)
    param.addcdiv_(m, v.sqrt().add_(eps), value=-lr * correction)


# -------------------------------------------------------------- batchnorm
# caffe BatchNorm: normalization only (affine lives in the Scale layer),
# so the output IS xhat — backward takes it instead of caching x.

def bn_forward_train(x, eps):
    xf = x.float()
    dims = [0] + list(range(2, xf.dim()))
    mean = xf.mean(dim=dims)
    var = xf.var(dim=dims, unbiased=False)
    invstd = (var + eps).rsqrt()
    shape = [1, -1] + [1] * (xf.dim() - 2)
    y = (xf - mean.reshape(shape)) * invstd.reshape(shape)
    return y.to(x.dtype), mean, var, invstd


def bn_forward_infer(x, mean, var, eps):
    invstd = (var.float() + eps).rsqrt()
    shape = [1, -1] + [1] * (x.dim() - 2)
    y = (x.float() - mean.float().reshape(shape)) * invstd.reshape(shape)
    return y.to(x.dtype), invstd


def bn_backward(xhat, dy, invstd, train):
    dyf, xh = dy.float(), xhat.float()
    shape = [1, -1] + [1] * (dy.dim() - 2)
    if not train:
        return (dyf * invstd.reshape(shape)).to(xhat.dtype)
    dims = [0] + list(range(2, dy.dim()))
    dmean = dyf.mean(dim=dims).reshape(shape)
    dvar = (dyf * xh).mean(dim=dims).reshape(shape)
    dx = (dyf - dmean - xh * dvar) * invstd.reshape(shape)
    return dx.to(xhat.dtype)
