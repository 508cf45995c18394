"""Per-layer gradient checks: our explicit backward vs torch autograd
through the same forward graph (reference impls are differentiable)."""

import pytest
import torch

from caffeonspark_amd.core.blob import Blob
from caffeonspark_amd.core.layers.base import create_layer
from caffeonspark_amd.proto import caffe_pb, text_format

torch.manual_seed(0)


class FakeNet:
    phase = caffe_pb.Phase.TRAIN
    dtype = torch.float32
    device = torch.device("cpu")
    generator = torch.Generator().manual_seed(1)


def make_layer(text):
    lp = text_format.parse(text, caffe_pb.LayerParameter)
    return create_layer(lp, FakeNet())


def run_grad_check(layer, bottoms, *, grad_bottoms=None, rtol=2e-3, atol=1e-4):
    """Forward with autograd enabled, then compare layer.backward outputs
    to autograd's vector-Jacobian products."""
    n_b = len(bottoms)
    grad_bottoms = grad_bottoms if grad_bottoms is not None else [True] * n_b
    bblobs = []
    leaves = []
    for t, g in zip(bottoms, grad_bottoms):
        t = t.detach().clone()
        if g:
            t.requires_grad_(True)
            leaves.append(t)
        b = Blob(t.shape)
        b.data = t
        bblobs.append(b)
    tblobs = [Blob([0]) for _ in range(max(1, len(layer.param.top)))]
    with torch.enable_grad():
        layer.setup(bblobs, tblobs)
        for p in layer.blobs:
            if p._lr_mult != 0:
                p.data.requires_grad_(True)
                leaves.append(p.data)
        layer.forward(bblobs, tblobs)
        y = tblobs[0].data
        dy = torch.randn_like(y)
        auto_grads = torch.autograd.grad(y, leaves, dy, allow_unused=True)
    # our backward
    tblobs[0].diff = dy
    for p in layer.blobs:
        p.data = p.data.detach()
        p.zero_diff()
    layer.backward(tblobs, grad_bottoms, bblobs)
    ours = []
    for b, g in zip(bblobs, grad_bottoms):
        if g:
            ours.append(b.diff)
    for p in layer.blobs:
        if p._lr_mult != 0:
            ours.append(p.diff)
    for i, (a, o) in enumerate(zip(auto_grads, ours)):
        if a is None or o is None:
            continue
        torch.testing.assert_close(o, a.to(o.dtype), rtol=rtol, atol=atol,
                                   msg=lambda m, i=i: f"grad {i}: {m}")


def test_conv_grads():
    layer = make_layer("""
        name: "c" type: "Convolution" bottom: "x" top: "y"
        convolution_param { num_output: 6 kernel_size: 3 stride: 2 pad: 1
          weight_filler { type: "xavier" } }""")
    run_grad_check(layer, [torch.randn(2, 4, 9, 9)])


def test_conv_group_grads():
    layer = make_layer("""
        name: "c" type: "Convolution" bottom: "x" top: "y"
        convolution_param { num_output: 8 kernel_size: 3 group: 2
          weight_filler { type: "gaussian" std: 0.1 } }""")
    run_grad_check(layer, [torch.randn(2, 4, 7, 7)])


def test_inner_product_grads():
    layer = make_layer("""
        name: "f" type: "InnerProduct" bottom: "x" top: "y"
        inner_product_param { num_output: 5
          weight_filler { type: "xavier" } }""")
    run_grad_check(layer, [torch.randn(3, 4, 2, 2)])


def test_relu_grads():
    layer = make_layer('name: "r" type: "ReLU" bottom: "x" top: "y"')
    run_grad_check(layer, [torch.randn(4, 8)])


def test_relu_leaky_grads():
    layer = make_layer("""name: "r" type: "ReLU" bottom: "x" top: "y"
                          relu_param { negative_slope: 0.1 }""")
    run_grad_check(layer, [torch.randn(4, 8)])


def test_sigmoid_tanh_grads():
    for t in ("Sigmoid", "TanH"):
        layer = make_layer(f'name: "a" type: "{t}" bottom: "x" top: "y"')
        run_grad_check(layer, [torch.randn(4, 8)])


def test_maxpool_grads():
    layer = make_layer("""name: "p" type: "Pooling" bottom: "x" top: "y"
        pooling_param { pool: MAX kernel_size: 3 stride: 2 }""")
    run_grad_check(layer, [torch.randn(2, 3, 9, 9)])


def test_avgpool_grads():
    layer = make_layer("""name: "p" type: "Pooling" bottom: "x" top: "y"
        pooling_param { pool: AVE kernel_size: 3 stride: 2 pad: 1 }""")
    run_grad_check(layer, [torch.randn(2, 3, 9, 9)])


def test_lrn_grads():
    layer = make_layer("""name: "n" type: "LRN" bottom: "x" top: "y"
        lrn_param { local_size: 5 alpha: 0.0001 beta: 0.75 }""")
    run_grad_check(layer, [torch.randn(2, 8, 5, 5)])


def test_softmax_grads():
    layer = make_layer('name: "s" type: "Softmax" bottom: "x" top: "y"')
    run_grad_check(layer, [torch.randn(4, 10)])


def test_dropout_grads():
    layer = make_layer("""name: "d" type: "Dropout" bottom: "x" top: "y"
        dropout_param { dropout_ratio: 0.4 }""")
    run_grad_check(layer, [torch.randn(6, 6)])


def test_embed_grads():
    layer = make_layer("""name: "e" type: "Embed" bottom: "i" top: "y"
        embed_param { num_output: 7 input_dim: 11
          weight_filler { type: "uniform" min: -1 max: 1 } }""")
    idx = torch.randint(0, 11, (5, 3)).float()
    run_grad_check(layer, [idx], grad_bottoms=[False])


def test_lstm_grads():
    layer = make_layer("""name: "l" type: "LSTM" bottom: "x" bottom: "cont"
        top: "h" recurrent_param { num_output: 6
          weight_filler { type: "uniform" min: -0.1 max: 0.1 } }""")
    T, N, D = 4, 3, 5
    x = torch.randn(T, N, D)
    cont = torch.ones(T, N)
    cont[0] = 0  # sequence start
    cont[2, 1] = 0  # mid-batch restart
    run_grad_check(layer, [x, cont], grad_bottoms=[True, False])


def test_concat_eltwise_slice_grads():
    layer = make_layer("""name: "c" type: "Concat" bottom: "a" bottom: "b"
        top: "y" concat_param { axis: 1 }""")
    run_grad_check(layer, [torch.randn(2, 3, 4, 4), torch.randn(2, 5, 4, 4)])
    layer = make_layer("""name: "e" type: "Eltwise" bottom: "a" bottom: "b"
        top: "y" eltwise_param { operation: SUM coeff: 1.0 coeff: -2.0 }""")
    run_grad_check(layer, [torch.randn(3, 4), torch.randn(3, 4)])
    layer = make_layer("""name: "e" type: "Eltwise" bottom: "a" bottom: "b"
        top: "y" eltwise_param { operation: MAX }""")
    run_grad_check(layer, [torch.randn(3, 4), torch.randn(3, 4)])
    layer = make_layer("""name: "e" type: "Eltwise" bottom: "a" bottom: "b"
        top: "y" eltwise_param { operation: PROD }""")
    run_grad_check(layer, [torch.randn(3, 4) + 2, torch.randn(3, 4) + 2])


def test_power_flatten_reshape_split_grads():
    layer = make_layer("""name: "p" type: "Power" bottom: "x" top: "y"
        power_param { power: 2.0 scale: 0.5 shift: 1.0 }""")
    run_grad_check(layer, [torch.rand(3, 4) + 0.5])
    layer = make_layer("""name: "f" type: "Flatten" bottom: "x" top: "y"
        flatten_param { axis: 1 }""")
    run_grad_check(layer, [torch.randn(2, 3, 4, 5)])
    layer = make_layer("""name: "r" type: "Reshape" bottom: "x" top: "y"
        reshape_param { shape { dim: 0 dim: -1 dim: 2 } }""")
    run_grad_check(layer, [torch.randn(2, 6, 2)])
    layer = make_layer('name: "s" type: "Split" bottom: "x" top: "a" top: "b"')
    run_grad_check(layer, [torch.randn(3, 4)])


def test_batchnorm_scale_grads():
    layer = make_layer('name: "bn" type: "BatchNorm" bottom: "x" top: "y"')
    run_grad_check(layer, [torch.randn(4, 3, 5, 5)], rtol=5e-3, atol=5e-4)
    layer = make_layer("""name: "sc" type: "Scale" bottom: "x" top: "y"
        scale_param { bias_term: true }""")
    run_grad_check(layer, [torch.randn(4, 3, 5, 5)])


def test_softmax_loss_matches_torch():
    layer = make_layer(
        'name: "l" type: "SoftmaxWithLoss" bottom: "x" bottom: "t" top: "loss"')
    x = torch.randn(8, 10)
    t = torch.randint(0, 10, (8,)).float()
    bx, bt = Blob(x.shape), Blob(t.shape)
    bx.data, bt.data = x, t
    top = [Blob([0])]
    layer.setup([bx, bt], top)
    layer.forward([bx, bt], top)
    expected = torch.nn.functional.cross_entropy(x, t.long())
    torch.testing.assert_close(top[0].data, expected, rtol=1e-5, atol=1e-6)
    # backward
    top[0].diff = torch.ones(())
    layer.backward(top, [True, False], [bx, bt])
    x2 = x.detach().requires_grad_(True)
    torch.nn.functional.cross_entropy(x2, t.long()).backward()
    torch.testing.assert_close(bx.diff, x2.grad, rtol=1e-5, atol=1e-6)


def test_softmax_loss_ignore_label():
    layer = make_layer("""name: "l" type: "SoftmaxWithLoss" bottom: "x"
        bottom: "t" top: "loss" loss_param { ignore_label: -1 }""")
    x = torch.randn(6, 5)
    t = torch.tensor([0, 1, -1, 2, -1, 4]).float()
    bx, bt = Blob(x.shape), Blob(t.shape)
    bx.data, bt.data = x, t
    top = [Blob([0])]
    layer.setup([bx, bt], top)
    layer.forward([bx, bt], top)
    expected = torch.nn.functional.cross_entropy(x, t.long(), ignore_index=-1)
    torch.testing.assert_close(top[0].data, expected, rtol=1e-5, atol=1e-6)


def test_accuracy():
    layer = make_layer(
        'name: "a" type: "Accuracy" bottom: "x" bottom: "t" top: "acc"')
    x = torch.tensor([[0.9, 0.1], [0.2, 0.8], [0.7, 0.3]])
    t = torch.tensor([0.0, 1.0, 1.0])
    bx, bt = Blob(x.shape), Blob(t.shape)
    bx.data, bt.data = x, t
    top = [Blob([0])]
    layer.setup([bx, bt], top)
    layer.forward([bx, bt], top)
    assert top[0].data.item() == pytest.approx(2.0 / 3.0)


def test_accuracy_top_k():
    """Accuracy with top_k > 1 (reference AccuracyLayer top-k argmax)."""
    layer = make_layer("""name: "a" type: "Accuracy" bottom: "x" bottom: "t"
        top: "acc" accuracy_param { top_k: 2 }""")
    x = torch.tensor([[0.1, 0.9, 0.5],     # top2 = {1, 2}
                      [0.8, 0.1, 0.7],     # top2 = {0, 2}
                      [0.3, 0.2, 0.1]])    # top2 = {0, 1}
    t_ = torch.tensor([2.0, 1.0, 0.0])     # hits: yes, no, yes
    from caffeonspark_amd.core.blob import Blob
    bx, bt = Blob(x.shape), Blob(t_.shape)
    bx.data, bt.data = x, t_
    top = [Blob([0])]
    layer.setup([bx, bt], top)
    layer.forward([bx, bt], top)
    assert float(top[0].data) == pytest.approx(2.0 / 3.0)


def test_inplace_bn_scale_relu_chain_grads():
    """The standard Caffe in-place BN->Scale->ReLU idiom: all three layers
    read and write ONE blob. Scale's forward overwrites the blob holding
    BN's xhat and ReLU overwrites Scale's output, so each layer must cache
    what its backward needs (Caffe: BatchNorm x_norm_, ScaleLayer's stashed
    in-place bottom). Verifies dx and dscale/dbias against autograd of the
    composed chain."""
    bn = make_layer('name: "bn" type: "BatchNorm" bottom: "x" top: "x"')
    sc = make_layer("""name: "sc" type: "Scale" bottom: "x" top: "x"
        scale_param { bias_term: true
          filler { type: "gaussian" std: 1.0 }
          bias_filler { type: "gaussian" std: 0.5 } }""")
    relu = make_layer('name: "r" type: "ReLU" bottom: "x" top: "x"')

    x0 = torch.randn(4, 3, 5, 5)
    x = x0.detach().clone().requires_grad_(True)
    blob = Blob(x.shape)
    blob.data = x
    chain = [bn, sc, relu]
    with torch.enable_grad():
        for layer in chain:
            layer.setup([blob], [blob])
        leaves = [x, sc.blobs[0].data.requires_grad_(True),
                  sc.blobs[1].data.requires_grad_(True)]
        for layer in chain:
            layer.forward([blob], [blob])
        y = blob.data
        dy = torch.randn_like(y)
        auto_dx, auto_dscale, auto_dbias = torch.autograd.grad(y, leaves, dy)
    for p in sc.blobs:
        p.data = p.data.detach()
        p.zero_diff()
    blob.diff = dy
    for layer in reversed(chain):
        layer.backward([blob], [True], [blob])
    torch.testing.assert_close(blob.diff, auto_dx, rtol=2e-3, atol=1e-4)
    torch.testing.assert_close(sc.blobs[0].diff, auto_dscale,
                               rtol=2e-3, atol=1e-4)
    torch.testing.assert_close(sc.blobs[1].diff, auto_dbias,
                               rtol=2e-3, atol=1e-4)


def test_slice_bias_silence_layers():
    # Slice: channel split with slice points (reference SliceLayer)
    layer = make_layer("""name: "sl" type: "Slice" bottom: "x"
        top: "a" top: "b" top: "c"
        slice_param { axis: 1 slice_point: 2 slice_point: 5 }""")
    run_grad_check(layer, [torch.randn(2, 8, 3, 3)])

    # Bias: learned per-channel bias broadcast over trailing axes
    layer = make_layer("""name: "bi" type: "Bias" bottom: "x" top: "y"
        bias_param { axis: 1 }""")
    run_grad_check(layer, [torch.randn(2, 4, 5, 5)])

    # Silence: consumes its bottom, produces nothing, zero gradient
    layer = make_layer('name: "si" type: "Silence" bottom: "x"')
    x = Blob([2, 3])
    x.data = torch.randn(2, 3)
    assert layer.forward([x], []) == 0.0
    layer.backward([], [True], [x])   # must not raise
