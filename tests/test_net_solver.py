"""Net graph building, phase/stage filtering, solver semantics,
snapshot/restore round-trip, and a LeNet convergence gate (the reference's
own integration bar: InterleaveTest asserts accuracy>0.8 — SURVEY.md §4)."""

import os

import pytest
import torch

from caffeonspark_amd.core import Net, Solver, net_from_prototxt, \
    solver_from_prototxt
from caffeonspark_amd.proto import caffe_pb, text_format

LENET_SOLVER = "caffeonspark_amd/models/lenet_memory_solver.prototxt"


def synth_batch(n, g, leak=True):
    """Synthetic separable 10-class data in MNIST shape: class-c samples
    carry a vertical stripe at column 2c."""
    x = torch.randn(n, 1, 28, 28, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    if leak:
        for i in range(n):
            x[i, 0, :, int(y[i]) * 2] += 3.0
    return x, y.float()


def test_phase_filtering():
    net = net_from_prototxt(
        "caffeonspark_amd/models/lenet_memory_train_test.prototxt",
        state=caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN))
    names = [l.name for l in net.layers]
    assert "accuracy" not in names  # TEST-only layer filtered
    assert names.count("data") == 1
    test_net = net_from_prototxt(
        "caffeonspark_amd/models/lenet_memory_train_test.prototxt",
        state=caffe_pb.NetState(phase=caffe_pb.Phase.TEST))
    assert "accuracy" in [l.name for l in test_net.layers]


def test_stage_filtering():
    text = """
    layer { name: "a" type: "DummyData" top: "a"
            dummy_data_param { shape { dim: 1 dim: 2 } } }
    layer { name: "b" type: "ReLU" bottom: "a" top: "b"
            include { stage: "s1" } }
    layer { name: "c" type: "ReLU" bottom: "a" top: "c"
            include { not_stage: "s1" } }
    """
    param = text_format.parse(text, caffe_pb.NetParameter)
    net = Net(param, caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN,
                                       stage=["s1"]))
    assert [l.name for l in net.layers] == ["a", "b"]
    net = Net(param, caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN))
    assert [l.name for l in net.layers] == ["a", "c"]


def test_diff_fanin_accumulation():
    """A blob consumed by two layers must receive the sum of both grads."""
    text = """
    layer { name: "x" type: "DummyData" top: "x"
            dummy_data_param { shape { dim: 2 dim: 3 }
                               data_filler { type: "gaussian" std: 1.0 } } }
    layer { name: "p1" type: "Power" bottom: "x" top: "y1"
            power_param { scale: 2.0 } loss_weight: 1.0 }
    layer { name: "p2" type: "Power" bottom: "x" top: "y2"
            power_param { scale: 3.0 } loss_weight: 1.0 }
    """
    param = text_format.parse(text, caffe_pb.NetParameter)
    param.force_backward = True
    net = Net(param, caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN))
    net.forward()
    net.backward()
    dx = net.blob_by_name("x").diff
    # d(sum(2x) + sum(3x))/dx = 5
    torch.testing.assert_close(dx, torch.full_like(dx, 5.0))


def test_inplace_chain_backward():
    text = """
    layer { name: "x" type: "DummyData" top: "x"
            dummy_data_param { shape { dim: 4 dim: 4 }
                               data_filler { type: "gaussian" std: 1.0 } } }
    layer { name: "ip" type: "InnerProduct" bottom: "x" top: "h"
            inner_product_param { num_output: 4
              weight_filler { type: "xavier" } } }
    layer { name: "r" type: "ReLU" bottom: "h" top: "h" }
    layer { name: "p" type: "Power" bottom: "h" top: "y"
            power_param { scale: 1.0 } loss_weight: 1.0 }
    """
    param = text_format.parse(text, caffe_pb.NetParameter)
    net = Net(param, caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN))
    net.forward()
    net.backward()
    ip = net.layer_by_name("ip")
    assert ip.blobs[0].diff is not None
    assert float(ip.blobs[0].diff.abs().sum()) > 0


def test_lr_policies():
    base = dict(base_lr=1.0, max_iter=100)
    cases = [
        (dict(lr_policy="fixed"), 50, 1.0),
        (dict(lr_policy="step", gamma=0.1, stepsize=10), 25, 0.01),
        (dict(lr_policy="inv", gamma=0.5, power=1.0), 2, 0.5),
        (dict(lr_policy="poly", power=1.0), 50, 0.5),
        (dict(lr_policy="exp", gamma=0.5), 3, 0.125),
        (dict(lr_policy="sigmoid", gamma=-1.0, stepsize=10), 10, 0.5),
    ]
    net_text = ('layer { name: "x" type: "DummyData" top: "x" '
                'dummy_data_param { shape { dim: 1 dim: 1 } } }')
    for kw, it, expect in cases:
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            **base, **kw)
        s = Solver(sp)
        s.iter = it
        assert s.get_lr() == pytest.approx(expect, rel=1e-6), kw

    # multistep: gamma applied at each stepvalue crossing (stateful)
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(net_text, caffe_pb.NetParameter),
        lr_policy="multistep", gamma=0.1, stepvalue=[5, 8], **base)
    s = Solver(sp)
    seen = []
    for it in range(10):
        s.iter = it
        seen.append(round(s.get_lr(), 6))
    assert seen[:5] == [1.0] * 5          # before first step
    assert seen[5:8] == [0.1, 0.1, 0.1]   # after stepvalue 5
    assert seen[8:] == [0.01, 0.01]       # after stepvalue 8


def test_sgd_update_matches_manual():
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse("""
          layer { name: "x" type: "DummyData" top: "x" top: "t"
                  dummy_data_param { shape { dim: 4 dim: 3 } shape { dim: 4 }
                                     data_filler { type: "gaussian" std: 1.0 } } }
          layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
                  inner_product_param { num_output: 2
                    weight_filler { type: "gaussian" std: 0.1 } } }
          layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
                  top: "loss" }
        """, caffe_pb.NetParameter),
        base_lr=0.1, momentum=0.9, weight_decay=0.01, lr_policy="fixed",
        max_iter=10, random_seed=3)
    s = Solver(sp)
    w = s.params[0]
    w0 = w.data.clone()
    s.net.forward_backward()
    g = w.diff.clone()
    expected = w0 - 0.1 * (g + 0.01 * w0)
    s.apply_update()
    torch.testing.assert_close(w.data, expected, rtol=1e-5, atol=1e-7)


@pytest.mark.parametrize("stype,kw", [
    ("SGD", dict(base_lr=0.2, momentum=0.9)),
    ("Nesterov", dict(base_lr=0.2, momentum=0.9)),
    ("AdaGrad", dict(base_lr=0.5, momentum=0.0)),
    ("RMSProp", dict(base_lr=0.05, momentum=0.0, rms_decay=0.95)),
    ("AdaDelta", dict(base_lr=1.0, momentum=0.95, delta=1e-6)),
    ("Adam", dict(base_lr=0.05, momentum=0.9, momentum2=0.999)),
])
def test_solver_family_descends(stype, kw):
    """Each solver type (reference SolverRegistry catalog, SURVEY.md
    §2.5) must actually optimize: loss on a fixed separable batch drops
    well below the random-guessing baseline."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 64 channels: 8 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 4
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss" }
    """
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(net_text, caffe_pb.NetParameter),
        lr_policy="fixed", max_iter=400, random_seed=7, type=stype,
        weight_decay=0.0, **kw)
    s = Solver(sp)
    g = torch.Generator().manual_seed(1)
    t = torch.randint(0, 4, (64,), generator=g).float()
    x = torch.randn(64, 8, 1, 1, generator=g) * 0.3
    x[torch.arange(64), t.long() * 2, 0, 0] += 3.0  # separable signal
    s.net.data_layers()[0].reset(x, t)
    first = s._step_one()
    last = s.step(300)
    assert last < 0.35, f"{stype}: loss {first:.3f} -> {last:.3f}"
    assert last < first * 0.5, f"{stype} barely descended"


def test_snapshot_restore_roundtrip(tmp_path):
    os.chdir(tmp_path)
    try:
        s = solver_from_prototxt(
            os.path.join(os.path.dirname(__file__), "..", LENET_SOLVER))
        g = torch.Generator().manual_seed(0)
        x, y = synth_batch(64, g)
        s.net.data_layers()[0].reset(x, y)
        s.step(3)
        model_file = s.snapshot()
        assert os.path.exists(model_file)
        assert os.path.exists(s.snapshot_filename("state"))
        w_before = [b.data.clone() for b in s.params]
        h_before = [h.clone() for h in s.history]

        s2 = solver_from_prototxt(
            os.path.join(os.path.dirname(__file__), "..", LENET_SOLVER))
        s2.restore(s.snapshot_filename("state"))
        assert s2.iter == 3
        for a, b in zip(w_before, [p.data for p in s2.params]):
            torch.testing.assert_close(a, b)
        for a, b in zip(h_before, s2.history):
            torch.testing.assert_close(a, b)
    finally:
        os.chdir(os.path.dirname(os.path.dirname(__file__)))


def test_lenet_converges_synthetic():
    """Convergence gate mirroring the reference's InterleaveTest bar."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    s = solver_from_prototxt(os.path.join(root, LENET_SOLVER))
    g = torch.Generator().manual_seed(42)
    for i in range(100):
        x, y = synth_batch(64, g)
        s.net.data_layers()[0].reset(x, y)
        s._step_one()
    # evaluate on fresh data with the TEST net (shared weights)
    test_net = s.test_nets[0]
    x, y = synth_batch(100, g)
    test_net.data_layers()[0].reset(x, y)
    test_net.forward()
    acc = float(test_net.blob_by_name("accuracy").data)
    loss = float(test_net.blob_by_name("loss").data)
    assert acc > 0.8, f"accuracy {acc}"
    assert loss < 0.5, f"loss {loss}"


def test_graph_step_cpu_fallback():
    """graph_step on CPU (or any non-capturable config) must silently run
    the eager step."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 8 channels: 4 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 3
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss" }
    """
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(net_text, caffe_pb.NetParameter),
        base_lr=0.1, lr_policy="fixed", max_iter=10, random_seed=2)
    s = Solver(sp)
    g = torch.Generator().manual_seed(0)
    x = torch.randn(8, 4, 1, 1, generator=g)
    t_ = torch.randint(0, 3, (8,), generator=g).float()
    s.net.data_layers()[0].reset(x, t_)
    l1 = s.graph_step()
    l2 = s.graph_step()
    assert s.iter == 2
    assert l1 > 0 and l2 > 0


def test_resume_with_explicit_weights(tmp_path):
    """-snapshot + -weights: the explicit weights win over the state's
    learned_net (reference setLearnedNet semantics)."""
    os.chdir(tmp_path)
    try:
        root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        s = solver_from_prototxt(os.path.join(root, LENET_SOLVER))
        g = torch.Generator().manual_seed(0)
        x, y = synth_batch(64, g)
        s.net.data_layers()[0].reset(x, y)
        s.step(3)
        s.snapshot()
        state3 = s.snapshot_filename("state")
        s.step(3)
        model6 = s.snapshot()
        w6 = s.flat_w.clone()

        s2 = solver_from_prototxt(os.path.join(root, LENET_SOLVER))
        s2.restore(state3)          # state from iter 3
        s2.load_weights(model6)     # explicit newer weights win
        assert s2.iter == 3
        torch.testing.assert_close(s2.flat_w, w6)
    finally:
        os.chdir(os.path.dirname(os.path.dirname(__file__)))


def test_finetune_then_train_updates_weights(tmp_path):
    """Regression: load_weights must keep param blobs as arena views so
    subsequent training actually moves the weights the net reads."""
    os.chdir(tmp_path)
    try:
        root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        s = solver_from_prototxt(os.path.join(root, LENET_SOLVER))
        g = torch.Generator().manual_seed(1)
        x, y = synth_batch(64, g)
        s.net.data_layers()[0].reset(x, y)
        s.step(2)
        model = s.snapshot()

        s2 = solver_from_prototxt(os.path.join(root, LENET_SOLVER))
        s2.load_weights(model)
        # blobs must still alias the arena
        assert s2.params[0].data.data_ptr() >= s2.flat_w.data_ptr()
        assert s2.params[0].data.data_ptr() < \
            s2.flat_w.data_ptr() + s2.flat_w.numel() * 4
        w0 = s2.flat_w.clone()
        s2.net.data_layers()[0].reset(x, y)
        loss_a = s2._step_one()
        assert not torch.equal(w0, s2.flat_w), "training froze after load"
        # and the net's own view moved too
        assert not torch.equal(
            w0.narrow(0, 0, s2.params[0].count),
            s2.params[0].data.reshape(-1))
    finally:
        os.chdir(os.path.dirname(os.path.dirname(__file__)))


def test_iter_size_accumulation():
    """iter_size=2 over two half-batches equals one update with the
    averaged gradient (caffe gradient-scaling rule, CaffeNet.cpp:620)."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 16 channels: 6 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 3
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss" }
    """
    g = torch.Generator().manual_seed(3)
    x = torch.randn(32, 6, 1, 1, generator=g)
    t_ = torch.randint(0, 3, (32,), generator=g).float()

    def make(iter_size):
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.5, momentum=0.0, weight_decay=0.0,
            lr_policy="fixed", max_iter=4, random_seed=11,
            iter_size=iter_size)
        return Solver(sp)

    s = make(2)
    s.net.data_layers()[0].reset(x, t_)
    w0 = s.flat_w.clone()
    s._step_one()                         # two micro-batches of 16
    dw_acc = s.flat_w - w0

    # manual: average of the two half-batch gradients
    s2 = make(1)
    grads = []
    for half in (slice(0, 16), slice(16, 32)):
        s2.net.zero_param_diffs()
        s2.flat_w.copy_(w0)
        s2.net.data_layers()[0].reset(x[half], t_[half])
        s2.net.forward_backward()
        grads.append(s2.flat_g.clone())
    expected = -0.5 * 0.5 * (grads[0] + grads[1])   # lr * avg grad
    torch.testing.assert_close(dw_acc, expected, rtol=1e-4, atol=1e-6)


def test_gradient_clipping_and_l1():
    """clip_gradients rescales by global L2 norm; L1 regularization adds
    sign(w)*decay (reference SGDSolver::ClipGradients / Regularize)."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 8 channels: 5 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 2
                weight_filler { type: "gaussian" std: 0.5 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss" }
    """
    g = torch.Generator().manual_seed(4)
    x = torch.randn(8, 5, 1, 1, generator=g) * 5
    t_ = torch.randint(0, 2, (8,), generator=g).float()

    def run(**kw):
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=1.0, momentum=0.0, lr_policy="fixed", max_iter=4,
            random_seed=9, **kw)
        s = Solver(sp)
        s.net.data_layers()[0].reset(x, t_)
        w0 = s.flat_w.clone()
        s.net.zero_param_diffs()
        s.net.forward_backward()
        grad = s.flat_g.clone()
        s.apply_update()
        return w0, grad, s

    # clipping: with a tiny threshold the applied step has that norm
    clip = 0.01
    w0, grad, s = run(weight_decay=0.0, clip_gradients=clip)
    step = w0 - s.flat_w
    norm = float(grad.norm())
    if norm > clip:
        assert abs(float(step.norm()) - clip) / clip < 1e-3

    # L1: update includes sign(w) * decay
    wd = 0.05
    w0, grad, s = run(weight_decay=wd, regularization_type="L1")
    expected = w0 - (grad + wd * torch.sign(w0))
    torch.testing.assert_close(s.flat_w, expected, rtol=1e-4, atol=1e-6)


def test_loss_weight_scaling():
    """Auxiliary heads (GoogLeNet loss1/loss2, loss_weight 0.3): total
    loss and gradients scale by the declared loss_weight."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 8 channels: 4 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 3
                weight_filler { type: "gaussian" std: 0.2 } } }
      layer { name: "lmain" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss_main" }
      layer { name: "laux" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss_aux" loss_weight: 0.3 }
    """
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(net_text, caffe_pb.NetParameter),
        base_lr=0.1, lr_policy="fixed", max_iter=4, random_seed=6)
    s = Solver(sp)
    g = torch.Generator().manual_seed(0)
    x = torch.randn(8, 4, 1, 1, generator=g)
    t_ = torch.randint(0, 3, (8,), generator=g).float()
    s.net.data_layers()[0].reset(x, t_)
    total = s.net.forward()
    lm = float(s.net.blob_by_name("loss_main").data)
    la = float(s.net.blob_by_name("loss_aux").data)
    assert total == pytest.approx(lm + 0.3 * la, rel=1e-5)
    # both heads share the same logits: grad = (1 + 0.3) * single-head
    s.net.zero_param_diffs()
    s.net.backward()
    g2 = s.flat_g.clone()
    assert float(g2.norm()) > 0
    # single-head comparison net
    sp1 = caffe_pb.SolverParameter(
        net_param=text_format.parse(
            net_text.replace(' loss_weight: 0.3', ' loss_weight: 0.0'),
            caffe_pb.NetParameter),
        base_lr=0.1, lr_policy="fixed", max_iter=4, random_seed=6)
    s1 = Solver(sp1)
    s1.net.data_layers()[0].reset(x, t_)
    s1.net.forward()
    s1.net.zero_param_diffs()
    s1.net.backward()
    torch.testing.assert_close(g2, 1.3 * s1.flat_g, rtol=1e-4, atol=1e-6)


def test_seed_determinism_and_loss_smoothing():
    """random_seed gives identical init; smoothed loss averages the last
    average_loss iterations (reference UpdateSmoothedLoss)."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 4 channels: 3 height: 1
                                  width: 1 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "y"
              inner_product_param { num_output: 2
                weight_filler { type: "gaussian" std: 0.3 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "y" bottom: "t"
              top: "loss" }
    """

    def make():
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.0, lr_policy="fixed", max_iter=9, random_seed=21,
            average_loss=3)
        return Solver(sp)

    a, b = make(), make()
    torch.testing.assert_close(a.flat_w, b.flat_w, rtol=0, atol=0)

    # lr 0: weights frozen, per-step losses vary only with data; smoothing
    # window = mean of the last 3 step losses
    g = torch.Generator().manual_seed(5)
    losses = []
    for _ in range(5):
        x = torch.randn(4, 3, 1, 1, generator=g)
        t_ = torch.randint(0, 2, (4,), generator=g).float()
        a.net.data_layers()[0].reset(x, t_)
        losses.append(a._step_one())
    expect = sum(losses[-3:]) / 3
    assert a.smoothed_loss == pytest.approx(expect, rel=1e-6)


def test_model_zoo_generator_in_sync():
    """The committed model-zoo prototxts match what generate.py emits
    (regeneration is reproducible and nothing was hand-edited).  The
    generator writes in place; git must see no resulting change."""
    import subprocess

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    models = os.path.join(root, "caffeonspark_amd", "models")
    import caffeonspark_amd.models.generate as gen
    gen.main()
    out = subprocess.run(["git", "diff", "--name-only", "--", models],
                         capture_output=True, text=True, cwd=root)
    assert out.stdout.strip() == "", out.stdout


def test_named_param_sharing():
    """caffe param { name } sharing: two layers alias one blob, its
    gradient accumulates both layers' contributions, and the solver
    counts it once (siamese-style)."""
    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 6 channels: 4 height: 1
                                  width: 1 } }
      layer { name: "ipa" type: "InnerProduct" bottom: "x" top: "a"
              param { name: "w_shared" } param { name: "b_shared" }
              inner_product_param { num_output: 4
                weight_filler { type: "gaussian" std: 0.2 } } }
      layer { name: "ipb" type: "InnerProduct" bottom: "a" top: "b"
              param { name: "w_shared" } param { name: "b_shared" }
              inner_product_param { num_output: 4 } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "b" bottom: "t"
              top: "loss" }
    """
    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(net_text, caffe_pb.NetParameter),
        base_lr=0.1, lr_policy="fixed", max_iter=4, random_seed=13)
    s = Solver(sp)
    ipa = next(l for l in s.net.layers if l.name == "ipa")
    ipb = next(l for l in s.net.layers if l.name == "ipb")
    assert ipa.blobs[0] is ipb.blobs[0]
    assert ipa.blobs[1] is ipb.blobs[1]
    # deduped in the solver arena
    assert len(s.params) == 2

    g = torch.Generator().manual_seed(2)
    x = torch.randn(6, 4, 1, 1, generator=g)
    t_ = torch.randint(0, 4, (6,), generator=g).float()
    s.net.data_layers()[0].reset(x, t_)
    s.net.zero_param_diffs()
    s.net.forward_backward()
    shared_grad = ipa.blobs[0].diff.clone()

    # autograd oracle for the weight-tied two-layer chain
    w = ipa.blobs[0].data.detach().clone().requires_grad_(True)
    bias = ipa.blobs[1].data.detach().clone().requires_grad_(True)
    xf = x.reshape(6, 4)
    h = torch.nn.functional.linear(xf, w, bias)
    out = torch.nn.functional.linear(h, w, bias)
    loss = torch.nn.functional.cross_entropy(out, t_.long())
    loss.backward()
    torch.testing.assert_close(shared_grad, w.grad, rtol=1e-4, atol=1e-6)


def test_dropout_reproducible_from_solver_seed():
    """solver random_seed must cover dropout masks (upstream Caffe seeds
    dropout through its global RNG): two same-seed solvers produce
    bit-identical weights after training steps through a Dropout net."""
    import torch

    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 8 channels: 1 height: 6
                                  width: 6 } }
      layer { name: "ip" type: "InnerProduct" bottom: "x" top: "h"
              inner_product_param { num_output: 16
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "drop" type: "Dropout" bottom: "h" top: "h"
              dropout_param { dropout_ratio: 0.5 } }
      layer { name: "ip2" type: "InnerProduct" bottom: "h" top: "z"
              inner_product_param { num_output: 3
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "z" bottom: "t"
              top: "loss" }
    """

    def run():
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.1, lr_policy="fixed", max_iter=5, random_seed=42,
            display=0)
        s = Solver(sp)
        g = torch.Generator().manual_seed(3)
        for _ in range(3):
            x = torch.randn(8, 1, 6, 6, generator=g)
            y = torch.randint(0, 3, (8,), generator=g).float()
            s.net.data_layers()[0].reset(x, y)
            s._step_one()
        return s.flat_w.clone()

    torch.manual_seed(999)   # global RNG noise must not matter
    w1 = run()
    torch.manual_seed(123)
    w2 = run()
    torch.testing.assert_close(w1, w2, rtol=0, atol=0)
