"""gfx950 kernel numerics: every HIP op vs the plain-torch fp32 reference.

Inputs are rounded to bf16 first, then the reference runs in fp32 on the
rounded values, so tolerances only cover accumulation-order and output-
rounding differences (guide G9: random asymmetric inputs catch transposed
layouts)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from caffeonspark_amd import ops
from caffeonspark_amd.ops import reference


def dev():
    return torch.device("cuda:0")


def bf(x):
    return x.to(torch.bfloat16)


def agree(got, want, rtol=0.02, atol=0.02):
    torch.testing.assert_close(got.float().cpu(), want.float().cpu(),
                               rtol=rtol, atol=atol)


@pytest.fixture(scope="module", autouse=True)
def _seed():
    torch.manual_seed(7)


# ------------------------------------------------------------------- GEMM

def _gemm_ref(a, b, ta, tb):
    af = a.float().t() if ta else a.float()
    bf_ = b.float().t() if tb else b.float()
    return af @ bf_.t()


@pytest.mark.parametrize("ta,tb", [(False, False), (False, True),
                                   (True, False), (True, True)])
@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 384, 512),
                                   (100, 96, 363), (130, 70, 40),
                                   (512, 1024, 768)])
def test_gemm_combos(ta, tb, m, n, k):
    ext = ops.native()
    a_shape = (k, m) if ta else (m, k)
    b_shape = (k, n) if tb else (n, k)
    a = bf(torch.randn(a_shape)).to(dev())
    b = bf(torch.randn(b_shape)).to(dev())
    c = torch.empty(m, n, dtype=torch.bfloat16, device=dev())
    ext.gemm(a, b, c, None, m, n, k, a.shape[1], b.shape[1], n,
             ta, tb, 0, 1, False, 1.0, m, n)
    torch.cuda.synchronize()
    want = _gemm_ref(a.cpu(), b.cpu(), ta, tb)
    agree(c, want, rtol=0.03, atol=0.05 * (k ** 0.5) * 0.02)


def test_gemm_bias_relu():
    ext = ops.native()
    m, n, k = 200, 130, 96
    a = bf(torch.randn(m, k)).to(dev())
    b = bf(torch.randn(n, k)).to(dev())
    bias = torch.randn(n).to(dev())
    c = torch.empty(m, n, dtype=torch.bfloat16, device=dev())
    ext.gemm(a, b, c, bias, m, n, k, k, k, n, False, False, 0, 1, True,
             1.0, m, n)
    torch.cuda.synchronize()
    want = torch.relu(_gemm_ref(a.cpu(), b.cpu(), False, False)
                      + bias.cpu().float())
    agree(c, want, rtol=0.03, atol=0.1)


def test_gemm_splitk_atomic():
    ext = ops.native()
    m, n, k = 96, 384, 8192
    a = bf(torch.randn(k, m) * 0.1).to(dev())   # trans A
    b = bf(torch.randn(k, n) * 0.1).to(dev())   # trans B
    c = torch.zeros(m, n, dtype=torch.float32, device=dev())
    ext.gemm(a, b, c, None, m, n, k, m, n, n, True, True, 2, 8, False,
             1.0, m, n)
    torch.cuda.synchronize()
    want = a.cpu().float().t() @ b.cpu().float()
    agree(c, want, rtol=0.03, atol=0.5)


# ------------------------------------------------------------------- conv

@pytest.mark.parametrize("case", [
    dict(n=4, c=3, h=33, w=33, k=32, r=11, stride=4, pad=0, groups=1),
    dict(n=2, c=16, h=15, w=15, k=32, r=5, stride=1, pad=2, groups=2),
    dict(n=2, c=24, h=13, w=13, k=48, r=3, stride=1, pad=1, groups=1),
    dict(n=2, c=32, h=9, w=9, k=16, r=1, stride=1, pad=0, groups=1),
    dict(n=2, c=32, h=8, w=8, k=16, r=1, stride=1, pad=0, groups=1),
    # implicit fwd/dw with the dcol+col2im dx fallback (stride 2)
    dict(n=2, c=16, h=15, w=15, k=24, r=3, stride=2, pad=1, groups=1),
    # implicit dx at maximal pad (pad' == 0 boundary)
    dict(n=2, c=8, h=12, w=12, k=16, r=3, stride=1, pad=2, groups=1),
    # Cg%8 != 0: materialized-col fallback for an 8-misaligned group
    dict(n=2, c=16, h=11, w=11, k=16, r=3, stride=1, pad=1, groups=4),
    # dilation through the implicit decode
    dict(n=2, c=16, h=15, w=15, k=16, r=3, stride=1, pad=2, groups=1,
         dil=2),
    # small-C implicit (contiguous-run staging) WITH padding: edge
    # slots take the per-element guard path
    dict(n=2, c=4, h=21, w=21, k=16, r=9, stride=2, pad=4, groups=1),
])
def test_conv_forward_backward(case):
    n, c, h, w = case["n"], case["c"], case["h"], case["w"]
    k, r, st, pd, g = (case["k"], case["r"], case["stride"], case["pad"],
                       case["groups"])
    dil = case.get("dil", 1)
    x = bf(torch.randn(n, c, h, w)).to(dev())
    wt = bf(torch.randn(k, c // g, r, r) * 0.1).to(dev()).float()
    b = torch.randn(k).to(dev()).float()
    ctx = {}
    y = ops.conv2d_forward(x, wt, b, (st, st), (pd, pd), (dil, dil), g,
                           ctx=ctx)
    x_cpu, w_cpu, b_cpu = x.float().cpu(), wt.float().cpu(), b.float().cpu()
    want = reference.conv2d_forward(x_cpu, w_cpu, b_cpu, (st, st), (pd, pd),
                                    (dil, dil), g)
    agree(y, want, rtol=0.05, atol=0.1)

    dy = bf(torch.randn_like(y.float())).to(dev())
    dx, dw, db = ops.conv2d_backward(x, wt, dy, (st, st), (pd, pd),
                                     (dil, dil), g, ctx=ctx)
    rdx, rdw, rdb = reference.conv2d_backward(
        x_cpu, w_cpu, dy.float().cpu(), (st, st), (pd, pd), (dil, dil), g)
    agree(dx, rdx, rtol=0.05, atol=0.15)
    agree(dw, rdw, rtol=0.05, atol=0.3)
    agree(db, rdb, rtol=0.05, atol=0.3)


# --------------------------------------------------------------------- fc

def test_fc_forward_backward():
    m, k, n = 64, 500, 77
    x = bf(torch.randn(m, k)).to(dev())
    w = bf(torch.randn(n, k) * 0.05).to(dev()).float()
    b = torch.randn(n).to(dev()).float()
    y = ops.fc_forward(x, w.to(torch.bfloat16), b.to(torch.bfloat16))
    want = reference.fc_forward(x.float().cpu(), w.cpu(), b.cpu())
    agree(y, want, rtol=0.03, atol=0.2)

    dy = bf(torch.randn(m, n)).to(dev())
    dx, dw, db = ops.fc_backward(x, w.to(torch.bfloat16), dy)
    rdx, rdw, rdb = reference.fc_backward(x.float().cpu(), w.cpu(),
                                          dy.float().cpu())
    agree(dx, rdx, rtol=0.03, atol=0.2)
    agree(dw, rdw, rtol=0.03, atol=0.2)
    agree(db, rdb, rtol=0.03, atol=0.2)


# ------------------------------------------------------------ elementwise

def test_relu_fwd_bwd():
    x = bf(torch.randn(3, 17, 9, 5)).to(dev())
    y = ops.relu_forward(x)
    agree(y, reference.relu_forward(x.float().cpu()), rtol=0, atol=0)
    dy = bf(torch.randn_like(x.float())).to(dev())
    dx = ops.relu_backward(y, dy)
    agree(dx, reference.relu_backward(y.float().cpu(), dy.float().cpu()))


def test_dropout():
    x = bf(torch.ones(100000)).to(dev())
    y, mask = ops.dropout_forward(x, 0.4)
    keep_frac = (mask.float() > 0).float().mean().item()
    assert abs(keep_frac - 0.6) < 0.02
    dy = bf(torch.randn_like(x.float())).to(dev())
    dx = ops.dropout_backward(mask, dy)
    agree(dx, dy.float().cpu() * mask.float().cpu())


# ---------------------------------------------------------------- pooling

def test_maxpool():
    x = bf(torch.randn(2, 16, 13, 13)).to(dev())
    y, pack = ops.maxpool_forward(x, (3, 3), (2, 2), (0, 0))
    ry, _ = reference.maxpool_forward(x.float().cpu(), (3, 3), (2, 2), (0, 0))
    agree(y, ry, rtol=0, atol=0)
    dy = bf(torch.randn_like(y.float())).to(dev())
    dx = ops.maxpool_backward(list(x.shape), pack, dy)
    # reference backward via autograd
    x2 = x.float().cpu().requires_grad_(True)
    with torch.enable_grad():
        yy = torch.nn.functional.max_pool2d(x2, 3, 2, 0, ceil_mode=True)
    yy.backward(dy.float().cpu())
    agree(dx, x2.grad)


def test_avgpool_with_pad():
    x = bf(torch.randn(2, 8, 14, 14)).to(dev())
    y = ops.avgpool_forward(x, (3, 3), (2, 2), (1, 1))
    ry = reference.avgpool_forward(x.float().cpu(), (3, 3), (2, 2), (1, 1))
    agree(y, ry, rtol=0.01, atol=0.01)
    dy = bf(torch.randn_like(y.float())).to(dev())
    dx = ops.avgpool_backward(x, (3, 3), (2, 2), (1, 1), dy)
    rdx = reference.avgpool_backward(x.float().cpu(), (3, 3), (2, 2), (1, 1),
                                     dy.float().cpu())
    agree(dx, rdx, rtol=0.02, atol=0.02)


def test_global_avgpool():
    x = bf(torch.randn(2, 32, 7, 7)).to(dev())
    y = ops.global_avgpool_forward(x)
    agree(y, reference.global_avgpool_forward(x.float().cpu()),
          rtol=0.01, atol=0.01)


# -------------------------------------------------------------------- LRN

def test_lrn():
    x = bf(torch.randn(2, 96, 9, 9)).to(dev())
    y, scale = ops.lrn_forward(x, 5, 1e-4, 0.75, 1.0)
    ry, rscale = reference.lrn_forward(x.float().cpu(), 5, 1e-4, 0.75, 1.0)
    agree(y, ry, rtol=0.02, atol=0.02)
    dy = bf(torch.randn_like(x.float())).to(dev())
    dx = ops.lrn_backward(x, y, scale, dy, 5, 1e-4, 0.75)
    rdx = reference.lrn_backward(x.float().cpu(), ry, rscale,
                                 dy.float().cpu(), 5, 1e-4, 0.75)
    agree(dx, rdx, rtol=0.03, atol=0.03)


# ----------------------------------------------------------- softmax loss

def test_softmax_loss():
    n, c = 64, 1000
    x = bf(torch.randn(n, c)).to(dev())
    lab = torch.randint(0, c, (n,)).float().to(dev())
    loss, pack, cnt = ops.softmax_loss_forward(x, lab, None)
    rloss, rprob, rcnt = reference.softmax_loss_forward(
        x.float().cpu(), lab.cpu(), None)
    assert cnt == rcnt
    assert abs(loss.item() - rloss.item()) / rloss.item() < 0.01
    dx = ops.softmax_loss_backward(pack, lab, None, 1.0 / cnt)
    rdx = reference.softmax_loss_backward(rprob, lab.cpu(), None, 1.0 / rcnt)
    agree(dx, rdx, rtol=0.02, atol=1e-4)


def test_softmax_loss_ignore():
    n, c = 32, 11
    x = bf(torch.randn(n, c)).to(dev())
    lab = torch.randint(0, c, (n,)).float()
    lab[::4] = -1
    labd = lab.to(dev())
    loss, pack, cnt = ops.softmax_loss_forward(x, labd, -1)
    rloss, rprob, rcnt = reference.softmax_loss_forward(
        x.float().cpu(), lab, -1)
    assert cnt == rcnt
    assert abs(loss.item() - rloss.item()) / max(rloss.item(), 1e-6) < 0.01


# ------------------------------------------------------------------ embed

def test_embed():
    v, e = 50, 32
    w = (torch.randn(v, e) * 0.1).to(dev())
    idx = torch.randint(0, v, (7, 3)).float().to(dev())
    y = ops.embed_forward(idx, w)
    want = reference.embed_forward(idx.cpu(), w.to(torch.bfloat16).float().cpu())
    agree(y, want, rtol=0.01, atol=0.01)
    dy = bf(torch.randn(7, 3, e)).to(dev())
    dw, db = ops.embed_backward(idx, dy, v)
    rdw, rdb = reference.embed_backward(idx.cpu(), dy.float().cpu(), v)
    agree(dw, rdw, rtol=0.02, atol=0.02)
    agree(db, rdb, rtol=0.02, atol=0.02)


# -------------------------------------------------------------- LSTM unit

def test_lstm_unit():
    n, h = 12, 20
    c_prev = torch.randn(n, h).to(dev())
    gates = bf(torch.randn(n, 4 * h)).to(dev())
    cont = torch.ones(n).to(dev())
    cont[::3] = 0
    c, hh, cache = ops.lstm_unit_forward(c_prev, gates, cont)
    rc, rh, rcache = reference.lstm_unit_forward(
        c_prev.cpu(), gates.float().cpu(), cont.cpu())
    agree(c, rc, rtol=0.02, atol=0.02)
    agree(hh, rh, rtol=0.02, atol=0.02)
    dc_next = torch.randn(n, h).to(dev())
    dh = bf(torch.randn(n, h)).to(dev())
    dcp, dg = ops.lstm_unit_backward(c_prev, cache, dc_next, dh)
    rdcp, rdg = reference.lstm_unit_backward(c_prev.cpu(), rcache,
                                             dc_next.cpu(),
                                             dh.float().cpu())
    agree(dcp, rdcp, rtol=0.03, atol=0.03)
    agree(dg, rdg.reshape(n, 4 * h), rtol=0.03, atol=0.03)


# -------------------------------------------------------------------- SGD

def test_sgd_update():
    n = 1003
    p = torch.randn(n).to(dev())
    g = torch.randn(n).to(dev())
    v = torch.randn(n).to(dev())
    p0, g0, v0 = p.cpu().clone(), g.cpu().clone(), v.cpu().clone()
    ops.sgd_update(p, g, v, 0.1, 0.9, 0.005)
    reference.sgd_update(p0, g0, v0, 0.1, 0.9, 0.005)
    agree(p, p0, rtol=1e-5, atol=1e-6)
    agree(v, v0, rtol=1e-5, atol=1e-6)


# --------------------------------------------------- end-to-end net parity

def test_lenet_step_parity():
    """One LeNet fwd/bwd on GPU bf16 vs CPU fp32 with identical weights."""
    import os

    from caffeonspark_amd.core import net_from_prototxt
    from caffeonspark_amd.proto import caffe_pb

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proto = os.path.join(root, "caffeonspark_amd", "models",
                         "lenet_memory_train_test.prototxt")
    state = caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN)
    # CPU net also in bf16 so the comparison isolates kernel correctness
    # (accumulation order), not fp32-vs-bf16 drift through the deep chain
    cpu_net = net_from_prototxt(proto, state=state, seed=3,
                                dtype=torch.bfloat16)
    gpu_net = net_from_prototxt(proto, state=state, device=dev(),
                                dtype=torch.bfloat16, seed=3)
    # copy weights cpu -> gpu
    for cl, gl in zip(cpu_net.layers, gpu_net.layers):
        for cb, gb in zip(cl.blobs, gl.blobs):
            gb.data.copy_(cb.data.to(gb.data.device))
    x = torch.randn(32, 1, 28, 28).to(torch.bfloat16)
    y = torch.randint(0, 10, (32,)).float()
    cpu_net.data_layers()[0].reset(x, y)
    gpu_net.data_layers()[0].reset(x.to(dev(), torch.bfloat16), y.to(dev()))
    closs = cpu_net.forward()
    gloss = gpu_net.forward()
    assert abs(closs - gloss) / max(abs(closs), 1e-6) < 0.05
    cpu_net.backward()
    gpu_net.backward()
    for cl, gl in zip(cpu_net.layers, gpu_net.layers):
        for cb, gb in zip(cl.blobs, gl.blobs):
            if cb.diff is None:
                continue
            cg, gg = cb.diff.float(), gb.diff.float().cpu()
            # relative L2 error: robust to bf16 noise on single elements
            rel = (cg - gg).norm() / cg.norm().clamp_min(1e-4)
            assert rel < 0.08, f"{cl.name} grad mismatch relL2={rel:.3f}"


def test_batchnorm_parity():
    """HIP NHWC BatchNorm (stats reduce + normalize + backward) vs the
    fp32 torch reference, train and global-stats modes."""
    from caffeonspark_amd.ops import gpu as g, reference as ref

    torch.manual_seed(11)
    x = bf(torch.randn(4, 32, 9, 9) * 2 + 0.5).to(dev())
    xc = x.float().cpu()
    y, mean, var, invstd = g.bn_forward_train(x, 1e-5)
    ry, rmean, rvar, rinvstd = ref.bn_forward_train(xc, 1e-5)
    agree(mean, rmean, rtol=0.02, atol=0.02)
    agree(var, rvar, rtol=0.03, atol=0.03)
    agree(y, ry, rtol=0.05, atol=0.05)
    dy = bf(torch.randn_like(xc)).to(dev())
    dx = g.bn_backward(y, dy, invstd, train=True)
    rdx = ref.bn_backward(ry, dy.float().cpu(), rinvstd, train=True)
    rel = (dx.float().cpu() - rdx.float()).norm() / rdx.float().norm()
    assert float(rel) < 0.05, f"bn dx relL2={float(rel):.4f}"
    # global-stats mode
    gm = torch.randn(32).to(dev())
    gv = torch.rand(32).to(dev()) + 0.5
    y2, inv2 = g.bn_forward_infer(x, gm, gv, 1e-5)
    ry2, rinv2 = ref.bn_forward_infer(xc, gm.cpu(), gv.cpu(), 1e-5)
    agree(y2, ry2, rtol=0.05, atol=0.05)
    dx2 = g.bn_backward(y2, dy, inv2, train=False)
    rdx2 = ref.bn_backward(ry2, dy.float().cpu(), rinv2, train=False)
    agree(dx2, rdx2, rtol=0.05, atol=0.05)


def test_winograd_conv_parity(monkeypatch):
    """F(2x2,3x3) Winograd path (COS_WINOGRAD=1) vs the direct im2col
    path: fwd/dx/dw/db must agree within bf16 training noise."""
    from caffeonspark_amd.ops import gpu as g

    torch.manual_seed(5)
    x = bf(torch.randn(2, 32, 13, 13)).to(dev())
    wt = (torch.randn(64, 32, 3, 3) * 0.1).to(dev()).float()
    b = torch.randn(64).to(dev()).float()
    dy = bf(torch.randn(2, 64, 13, 13)).to(dev())
    outs = {}
    for flag in ("1", "0"):
        monkeypatch.setenv("COS_WINOGRAD", flag)
        ctx = {}
        y = g.conv2d_forward(x, wt, b, (1, 1), (1, 1), (1, 1), 1, ctx=ctx)
        assert ctx.get("wino", False) == (flag == "1")
        outs[flag] = (y,) + g.conv2d_backward(x, wt, dy, (1, 1), (1, 1),
                                              (1, 1), 1, ctx=ctx)
    for name, a, c in zip(("y", "dx", "dw", "db"), outs["1"], outs["0"]):
        rel = (a.float() - c.float()).norm() / c.float().norm().clamp_min(1e-5)
        assert float(rel) < 0.02, f"winograd {name} relL2={float(rel):.4f}"


def test_graph_step_parity():
    """hipGraph-captured solver steps == eager steps (same seed, same
    data).  graph_step's capture performs 2 uncounted warmup steps, so
    1 graph_step == 3 eager steps; 2 more replays -> 5 total."""
    import os

    from caffeonspark_amd.core import solver_from_prototxt

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proto = os.path.join(root, "caffeonspark_amd", "models",
                         "lenet_memory_solver.prototxt")
    torch.manual_seed(7)
    x = torch.randn(32, 1, 28, 28).to(dev(), torch.bfloat16)
    y = torch.randint(0, 10, (32,)).float().to(dev())

    def make():
        s = solver_from_prototxt(proto, device=dev(), dtype=torch.bfloat16)
        s.param.display = 0
        dl = s.net.data_layers()[0]
        dl.batch_size = 32
        dl.reset(x, y)
        return s

    eager, graph = make(), make()
    for _ in range(5):
        eager._step_one()
    graph.graph_step()            # capture: 2 warmup + 1 replay
    graph.graph_step()
    graph.graph_step()
    assert graph.iter == 5
    torch.cuda.synchronize()
    rel = (eager.flat_w - graph.flat_w).norm() / \
        eager.flat_w.norm().clamp_min(1e-4)
    assert float(rel) < 1e-3, f"graph vs eager weight drift relL2={rel}"


def test_inception_block_parity():
    """GoogLeNet inception wiring (1x1 bypass + concat) GPU vs CPU."""
    import os

    from caffeonspark_amd.core import Net
    from caffeonspark_amd.proto import caffe_pb, text_format

    text = """
    layer { name: "data" type: "DummyData" top: "data"
            dummy_data_param { shape { dim: 2 dim: 32 dim: 9 dim: 9 }
                               data_filler { type: "gaussian" std: 1.0 } } }
    layer { name: "b1" type: "Convolution" bottom: "data" top: "b1"
            convolution_param { num_output: 16 kernel_size: 1
              weight_filler { type: "xavier" } } }
    layer { name: "relu_b1" type: "ReLU" bottom: "b1" top: "b1" }
    layer { name: "b3r" type: "Convolution" bottom: "data" top: "b3r"
            convolution_param { num_output: 8 kernel_size: 1
              weight_filler { type: "xavier" } } }
    layer { name: "b3" type: "Convolution" bottom: "b3r" top: "b3"
            convolution_param { num_output: 16 kernel_size: 3 pad: 1
              weight_filler { type: "xavier" } } }
    layer { name: "pool" type: "Pooling" bottom: "data" top: "pool"
            pooling_param { pool: MAX kernel_size: 3 stride: 1 pad: 1 } }
    layer { name: "bp" type: "Convolution" bottom: "pool" top: "bp"
            convolution_param { num_output: 8 kernel_size: 1
              weight_filler { type: "xavier" } } }
    layer { name: "cat" type: "Concat" bottom: "b1" bottom: "b3"
            bottom: "bp" top: "cat" concat_param { axis: 1 } }
    layer { name: "gap" type: "Pooling" bottom: "cat" top: "gap"
            pooling_param { pool: AVE global_pooling: true } }
    layer { name: "fc" type: "InnerProduct" bottom: "gap" top: "fc"
            inner_product_param { num_output: 5
              weight_filler { type: "xavier" } } }
    layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "lab"
            top: "loss" }
    layer { name: "lab" type: "DummyData" top: "lab"
            dummy_data_param { shape { dim: 2 } } }
    """
    # reorder: lab before loss
    param = text_format.parse(text, caffe_pb.NetParameter)
    layers = list(param.layer)
    lab = [l for l in layers if l.name == "lab"][0]
    layers.remove(lab)
    layers.insert(0, lab)
    param.layer = layers
    state = caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN)
    cpu = Net(param, state, seed=9, dtype=torch.bfloat16)
    gpu = Net(param, state, seed=9, device=dev(), dtype=torch.bfloat16)
    for cl, gl in zip(cpu.layers, gpu.layers):
        for cb, gb in zip(cl.blobs, gl.blobs):
            gb.data.copy_(cb.data.to(gb.data.device))
    closs = cpu.forward()
    gloss = gpu.forward()
    assert abs(closs - gloss) / max(abs(closs), 1e-5) < 0.05
    cpu.backward()
    gpu.backward()
    for cl, gl in zip(cpu.layers, gpu.layers):
        for cb, gb in zip(cl.blobs, gl.blobs):
            if cb.diff is None:
                continue
            rel = (cb.diff.float() - gb.diff.float().cpu()).norm() / \
                cb.diff.float().norm().clamp_min(1e-4)
            assert rel < 0.1, f"{cl.name} relL2={rel:.3f}"


def test_lstm_layer_gpu_parity():
    """Full LSTM layer (time-major, cont gating, static input) GPU vs CPU."""
    from caffeonspark_amd.core.layers.base import create_layer
    from caffeonspark_amd.core.blob import Blob
    from caffeonspark_amd.proto import caffe_pb, text_format

    text = '''name: "l" type: "LSTM" bottom: "x" bottom: "cont"
        top: "h" recurrent_param { num_output: 16
          weight_filler { type: "uniform" min: -0.1 max: 0.1 } }'''
    T, N, D = 5, 4, 12

    class FN:
        phase = caffe_pb.Phase.TRAIN
        dtype = torch.float32
        device = torch.device("cpu")
        generator = torch.Generator().manual_seed(2)

    class FNG:
        phase = caffe_pb.Phase.TRAIN
        dtype = torch.bfloat16
        device = dev()
        generator = torch.Generator().manual_seed(2)

    lp = text_format.parse(text, caffe_pb.LayerParameter)
    cl = create_layer(lp, FN())
    gl = create_layer(lp, FNG())
    x = torch.randn(T, N, D)
    cont = torch.ones(T, N)
    cont[0] = 0
    cont[3, 2] = 0

    def run(layer, xx, cc, dtype, device):
        bx, bc = Blob(xx.shape), Blob(cc.shape)
        bx.data = xx.to(device, dtype)
        bc.data = cc.to(device, dtype)
        top = [Blob([0])]
        layer.setup([bx, bc], top)
        return layer, [bx, bc], top

    cl, cbot, ctop = run(cl, x, cont, torch.float32, torch.device("cpu"))
    gl, gbot, gtop = run(gl, x, cont, torch.bfloat16, dev())
    for cb, gb in zip(cl.blobs, gl.blobs):
        gb.data.copy_(cb.data.to(gb.data.device))
    cl.forward(cbot, ctop)
    gl.forward(gbot, gtop)
    agree(gtop[0].data, ctop[0].data, rtol=0.05, atol=0.03)
    dy = torch.randn(T, N, 16)
    ctop[0].diff = dy
    gtop[0].diff = dy.to(dev(), torch.bfloat16)
    cl.backward(ctop, [True, False], cbot)
    gl.backward(gtop, [True, False], gbot)
    agree(gbot[0].diff, cbot[0].diff, rtol=0.08, atol=0.05)
    for i, (cb, gb) in enumerate(zip(cl.blobs, gl.blobs)):
        rel = (cb.diff.float() - gb.diff.float().cpu()).norm() / \
            cb.diff.float().norm().clamp_min(1e-4)
        assert rel < 0.1, f"param {i} relL2={rel:.3f}"


def test_fp32_gpu_path_parity():
    """The declared fp32 GPU path (-dtype fp32): activation ops route to
    the reference torch impls on ROCm (rocBLAS/MIOpen), reproducing fp32
    Caffe numerics on the GPU.  One AlexNet-class fwd/bwd + SGD step on
    GPU fp32 must closely match CPU fp32."""
    import os

    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sp_file = os.path.join(root, "caffeonspark_amd", "models",
                           "lenet_memory_solver.prototxt")
    sp = text_format.parse_file(sp_file, caffe_pb.SolverParameter)
    sp.random_seed = 9
    sp.display = 0

    def run(device):
        s = Solver(text_format.parse(text_format.dumps(sp),
                                     caffe_pb.SolverParameter),
                   device=device, dtype=torch.float32,
                   proto_dir=os.path.dirname(sp_file))
        g = torch.Generator().manual_seed(21)
        for _ in range(3):
            x = torch.randn(32, 1, 28, 28, generator=g)
            y = torch.randint(0, 10, (32,), generator=g).float()
            s.net.data_layers()[0].reset(x.to(device), y.to(device))
            s._step_one()
        return s.flat_w.cpu()

    w_cpu = run(torch.device("cpu"))
    w_gpu = run(dev())
    rel = (w_cpu - w_gpu).norm() / w_cpu.norm().clamp_min(1e-6)
    # measured ~4e-4 after 3 steps: pure fp32 accumulation-order noise
    # (rocBLAS/MIOpen vs CPU BLAS reduction order) amplified through the
    # updates; the bf16 kernel path lands >1e-2 on the same check, so
    # 2e-3 separates "fp32 numerics" from "bf16 numerics" cleanly
    assert rel < 2e-3, f"fp32 GPU path diverged from CPU fp32: relL2={rel}"


def test_fp32_gpu_dropout_lrn_ops():
    """fp32 CUDA tensors through dispatch: dropout (seeded CPU generator
    serving a CUDA tensor) and LRN run the reference impls exactly."""
    x = torch.randn(8, 16, 7, 7, device=dev())
    y, scale = ops.lrn_forward(x, 5, 1e-4, 0.75, 1.0)
    yr, sr = reference.lrn_forward(x.cpu(), 5, 1e-4, 0.75, 1.0)
    torch.testing.assert_close(y.cpu(), yr, rtol=1e-5, atol=1e-6)

    gen = torch.Generator().manual_seed(5)
    out1, m1 = ops.dropout_forward(x, 0.5, generator=gen)
    gen2 = torch.Generator().manual_seed(5)
    out2, m2 = ops.dropout_forward(x, 0.5, generator=gen2)
    torch.testing.assert_close(m1.cpu(), m2.cpu(), rtol=0, atol=0)


def test_lstm_persistent_vs_loop_path(monkeypatch):
    """The persistent whole-sequence LSTM kernel (grid-resident,
    slice-owned) against the per-step loop path: identical gate math,
    so h/c/act and the backward dxg/dwhc must agree tightly."""
    from caffeonspark_amd.ops import gpu as g

    T, N, H = 7, 48, 64        # H=64 -> 4 blocks: persistent-eligible
    H4 = 4 * H
    torch.manual_seed(3)
    whc = (torch.randn(H4, H) * 0.2).to(torch.bfloat16).to(dev())
    xg = (torch.randn(T, N, H4) * 0.5).to(torch.bfloat16).to(dev())
    cont = torch.ones(T, N)
    cont[0] = 0
    cont[3, :5] = 0            # mid-sequence resets exercise the gating
    cont = cont.to(torch.bfloat16).to(dev())
    dy = torch.randn(T, N, H).to(torch.bfloat16).to(dev())

    def run(persist):
        monkeypatch.setenv("COS_LSTM_PERSIST", "1" if persist else "0")
        h, cache = g.lstm_seq_forward(xg, whc, cont)
        dxg, dwhc = g.lstm_seq_backward(dy, whc, cache)
        torch.cuda.synchronize()
        return (h.float().cpu(), cache[2].cpu(), dxg.float().cpu(),
                dwhc.cpu())

    h_p, c_p, dxg_p, dwhc_p = run(True)
    h_l, c_l, dxg_l, dwhc_l = run(False)
    assert not torch.isnan(h_p).any()
    torch.testing.assert_close(h_p, h_l, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(c_p, c_l, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(dxg_p, dxg_l, rtol=3e-2, atol=3e-2)
    rel = (dwhc_p - dwhc_l).norm() / dwhc_l.norm().clamp_min(1e-6)
    assert rel < 2e-2, f"dwhc relL2={rel}"


def test_gpu_iter_size_accumulation():
    """iter_size=2 gradient accumulation on the GPU path: the fused
    dw-to-arena write (micro-batch 1, virgin overwrite) must compose
    with the accumulate path (micro-batch 2) — equals iter_size=1 over
    the concatenated batch within bf16 tolerance."""
    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 16 channels: 8 height: 9
                                  width: 9 } }
      layer { name: "c" type: "Convolution" bottom: "x" top: "y"
              convolution_param { num_output: 16 kernel_size: 3 pad: 1
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "r" type: "ReLU" bottom: "y" top: "y" }
      layer { name: "ip" type: "InnerProduct" bottom: "y" top: "z"
              inner_product_param { num_output: 5
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "z" bottom: "t"
              top: "loss" }
    """
    torch.manual_seed(11)
    xs = [torch.randn(16, 8, 9, 9).to(torch.bfloat16) for _ in range(2)]
    ys = [torch.randint(0, 5, (16,)).float() for _ in range(2)]

    def run(iter_size):
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.1, momentum=0.9, lr_policy="fixed", max_iter=4,
            random_seed=3, display=0, iter_size=iter_size)
        s = Solver(sp, device=dev(), dtype=torch.bfloat16)
        dl = s.net.data_layers()[0]
        if iter_size == 2:
            feed = iter([(xs[0], ys[0]), (xs[1], ys[1])] * 4)
            orig_forward = s.net.forward

            def forward(*a, **k):
                x, y = next(feed)
                dl.reset(x.to(dev()), y.to(dev()))
                return orig_forward(*a, **k)
            s.net.forward = forward
            s._step_one()
        else:
            dl.batch_size = 32
            dl.reset(torch.cat(xs).to(dev()), torch.cat(ys).to(dev()))
            s._step_one()
        return s.flat_w.cpu()

    w2 = run(2)
    w1 = run(1)
    # iter_size=2 averages two half-batch losses == full-batch mean
    rel = (w2 - w1).norm() / w1.norm().clamp_min(1e-6)
    assert rel < 5e-3, f"iter_size accumulation diverged: relL2={rel}"


@pytest.mark.parametrize("stype", ["Nesterov", "Adam"])
def test_fused_solver_updates_gpu(stype):
    """Fused whole-arena Nesterov/Adam kernels (round 2: non-SGD updates
    previously ran through per-blob torch glue) vs the CPU reference
    solver on identical grads."""
    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 8 channels: 4 height: 7
                                  width: 7 } }
      layer { name: "c" type: "Convolution" bottom: "x" top: "y"
              param { lr_mult: 1 } param { lr_mult: 2 decay_mult: 0 }
              convolution_param { num_output: 8 kernel_size: 3
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "ip" type: "InnerProduct" bottom: "y" top: "z"
              inner_product_param { num_output: 4
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "z" bottom: "t"
              top: "loss" }
    """

    def make(device, dtype):
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.05, momentum=0.9, momentum2=0.999, delta=1e-8,
            weight_decay=0.001, lr_policy="fixed", max_iter=8,
            random_seed=6, display=0, type=stype)
        return Solver(sp, device=device, dtype=dtype)

    g = torch.Generator().manual_seed(77)

    cpu = make(torch.device("cpu"), torch.float32)
    gpu = make(dev(), torch.bfloat16)
    gpu.flat_w.copy_(cpu.flat_w.to(dev()))
    # identical fp32 gradients into apply_update isolates the fused
    # kernel math from bf16 forward/backward noise
    for step in range(4):
        grads = torch.randn(int(cpu.flat_g.numel()), generator=g) * 0.1
        cpu.flat_g.copy_(grads)
        gpu.flat_g.copy_(grads.to(dev()))
        cpu.apply_update()
        gpu.apply_update()
        cpu.iter += 1
        gpu.iter += 1
    rel = (cpu.flat_w - gpu.flat_w.cpu()).norm() / \
        cpu.flat_w.norm().clamp_min(1e-6)
    assert rel < 1e-5, f"{stype}: relL2={rel}"


@pytest.mark.gpu
def test_bf16_shadow_coherence_through_updates_and_load(tmp_path):
    """The fused update kernels write the bf16 shadow from registers and
    refresh() then skips the whole-arena cast; weight loads must resync
    explicitly.  Verify the shadow tracks flat_w through steps, snapshot
    restore, and further steps."""
    import os

    from caffeonspark_amd.core.solver import solver_from_prototxt

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proto = os.path.join(root, "caffeonspark_amd", "models",
                         "lenet_memory_solver.prototxt")
    s = solver_from_prototxt(proto, device=dev(), dtype=torch.bfloat16)
    s.param.snapshot_prefix = str(tmp_path / "snap")
    x = torch.randn(32, 1, 28, 28).to(dev(), torch.bfloat16)
    y = torch.randint(0, 10, (32,)).float().to(dev())
    s.net.data_layers()[0].reset(x, y)

    def shadow_ok(sv):
        torch.cuda.synchronize()
        want = sv.flat_w.to(torch.bfloat16)
        assert torch.equal(sv.flat_wb, want), "bf16 shadow diverged"

    s.step(3)
    s.net.forward()          # refresh runs (skips the cast once synced)
    shadow_ok(s)

    model = s.snapshot()
    s.step(2)
    s.load_weights(model)    # out-of-band arena write -> resync path
    shadow_ok(s)
    s.step(2)
    s.net.forward()
    shadow_ok(s)
