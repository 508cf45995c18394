"""Generate the model-zoo prototxts (AlexNet/CaffeNet, GoogLeNet, CIFAR10-quick)
programmatically through our proto API.

Run:  python -m caffeonspark_amd.models.generate
Writes the .prototxt files next to this module.  The architectures are the
standard published ones (Krizhevsky et al. 2012 as BVLC CaffeNet; Szegedy et
al. 2014 GoogLeNet; Caffe cifar10_quick example) — the same model families
the reference ships configs for (reference data/bvlc_reference_net.prototxt
etc.), re-authored here.
"""

from __future__ import annotations

import os

from ..proto import caffe_pb as pb
from ..proto import text_format

HERE = os.path.dirname(os.path.abspath(__file__))


def _filler(**kw):
    return pb.FillerParameter(**kw)


def layer(net, name, type, bottom=(), top=(), phase=None, **params):
    lp = pb.LayerParameter(name=name, type=type,
                           bottom=list(bottom), top=list(top) or [name])
    if phase is not None:
        lp.include = [pb.NetStateRule(phase=phase)]
    for k, v in params.items():
        setattr(lp, k, v)
    net.layer.append(lp)
    return lp


def lr_params(w_lr=1.0, b_lr=2.0, w_decay=1.0, b_decay=0.0):
    return [pb.ParamSpec(lr_mult=w_lr, decay_mult=w_decay),
            pb.ParamSpec(lr_mult=b_lr, decay_mult=b_decay)]


def conv(net, name, bottom, nout, k, *, stride=1, pad=0, group=1,
         std=0.01, bias=0.0, relu=True, w_lr=1.0):
    lp = layer(net, name, "Convolution", [bottom], [name],
               convolution_param=pb.ConvolutionParameter(
                   num_output=nout, kernel_size=[k], stride=[stride],
                   pad=[pad] if pad else [], group=group,
                   weight_filler=_filler(type="gaussian", std=std),
                   bias_filler=_filler(type="constant", value=bias)))
    lp.param = lr_params(w_lr, 2 * w_lr)
    if relu:
        layer(net, "relu_" + name, "ReLU", [name], [name])
    return name


def fc(net, name, bottom, nout, *, std=0.005, bias=0.1, relu=True,
       dropout=None):
    lp = layer(net, name, "InnerProduct", [bottom], [name],
               inner_product_param=pb.InnerProductParameter(
                   num_output=nout,
                   weight_filler=_filler(type="gaussian", std=std),
                   bias_filler=_filler(type="constant", value=bias)))
    lp.param = lr_params()
    if relu:
        layer(net, "relu_" + name, "ReLU", [name], [name])
    if dropout:
        layer(net, "drop_" + name, "Dropout", [name], [name],
              dropout_param=pb.DropoutParameter(dropout_ratio=dropout))
    return name


def maxpool(net, name, bottom, k, stride, pad=0):
    layer(net, name, "Pooling", [bottom], [name],
          pooling_param=pb.PoolingParameter(
              pool=pb.PoolingParameter.PoolMethod.MAX, kernel_size=k,
              stride=stride, pad=pad))
    return name


def avepool(net, name, bottom, k, stride, pad=0):
    layer(net, name, "Pooling", [bottom], [name],
          pooling_param=pb.PoolingParameter(
              pool=pb.PoolingParameter.PoolMethod.AVE, kernel_size=k,
              stride=stride, pad=pad))
    return name


def lrn(net, name, bottom):
    layer(net, name, "LRN", [bottom], [name],
          lrn_param=pb.LRNParameter(local_size=5, alpha=0.0001, beta=0.75))
    return name


def memory_data(net, batch_train, batch_test, c, h, w, source="synthetic"):
    for phase, bs in ((pb.Phase.TRAIN, batch_train), (pb.Phase.TEST, batch_test)):
        lp = layer(net, "data", "MemoryData", [], ["data", "label"],
                   phase=phase,
                   memory_data_param=pb.MemoryDataParameter(
                       batch_size=bs, channels=c, height=h, width=w,
                       source=source, share_in_parallel=False))
        lp.source_class = "com.yahoo.ml.caffe.ImageDataFrame"


def alexnet(batch=256) -> pb.NetParameter:
    """BVLC CaffeNet (AlexNet-class: 5 conv + LRN + dropout), 227x227."""
    net = pb.NetParameter(name="CaffeNet")
    memory_data(net, batch, 50, 3, 227, 227)
    x = conv(net, "conv1", "data", 96, 11, stride=4, std=0.01)
    x = maxpool(net, "pool1", x, 3, 2)
    x = lrn(net, "norm1", x)
    x = conv(net, "conv2", x, 256, 5, pad=2, group=2, bias=1.0)
    x = maxpool(net, "pool2", x, 3, 2)
    x = lrn(net, "norm2", x)
    x = conv(net, "conv3", x, 384, 3, pad=1)
    x = conv(net, "conv4", x, 384, 3, pad=1, group=2, bias=1.0)
    x = conv(net, "conv5", x, 256, 3, pad=1, group=2, bias=1.0)
    x = maxpool(net, "pool5", x, 3, 2)
    x = fc(net, "fc6", x, 4096, dropout=0.5)
    x = fc(net, "fc7", x, 4096, dropout=0.5)
    x = fc(net, "fc8", x, 1000, std=0.01, bias=0.0, relu=False)
    layer(net, "accuracy", "Accuracy", [x, "label"], ["accuracy"],
          phase=pb.Phase.TEST)
    layer(net, "loss", "SoftmaxWithLoss", [x, "label"], ["loss"])
    return net


def cifar10_quick() -> pb.NetParameter:
    net = pb.NetParameter(name="CIFAR10_quick")
    memory_data(net, 100, 100, 3, 32, 32)
    lp = layer(net, "conv1", "Convolution", ["data"], ["conv1"],
               convolution_param=pb.ConvolutionParameter(
                   num_output=32, kernel_size=[5], pad=[2], stride=[1],
                   weight_filler=_filler(type="gaussian", std=0.0001),
                   bias_filler=_filler(type="constant")))
    lp.param = lr_params()
    x = maxpool(net, "pool1", "conv1", 3, 2)
    layer(net, "relu1", "ReLU", [x], [x])
    x = conv(net, "conv2", x, 32, 5, pad=2, std=0.01)
    x = avepool(net, "pool2", x, 3, 2)
    x = conv(net, "conv3", x, 64, 5, pad=2, std=0.01)
    x = avepool(net, "pool3", x, 3, 2)
    x = fc(net, "ip1", x, 64, std=0.1, bias=0.0, relu=False)
    x = fc(net, "ip2", x, 10, std=0.1, bias=0.0, relu=False)
    layer(net, "accuracy", "Accuracy", [x, "label"], ["accuracy"],
          phase=pb.Phase.TEST)
    layer(net, "loss", "SoftmaxWithLoss", [x, "label"], ["loss"])
    return net


def inception(net, name, bottom, n1, n3r, n3, n5r, n5, npool):
    """GoogLeNet inception module."""
    b1 = conv(net, f"{name}/1x1", bottom, n1, 1, std=0.03)
    b3r = conv(net, f"{name}/3x3_reduce", bottom, n3r, 1, std=0.09)
    b3 = conv(net, f"{name}/3x3", b3r, n3, 3, pad=1, std=0.03)
    b5r = conv(net, f"{name}/5x5_reduce", bottom, n5r, 1, std=0.2)
    b5 = conv(net, f"{name}/5x5", b5r, n5, 5, pad=2, std=0.03)
    bp = maxpool(net, f"{name}/pool", bottom, 3, 1, pad=1)
    bpp = conv(net, f"{name}/pool_proj", bp, npool, 1, std=0.1)
    out = f"{name}/output"
    layer(net, out, "Concat", [b1, b3, b5, bpp], [out])
    return out


def _aux_head(net, name, bottom):
    """GoogLeNet auxiliary classifier (training-only, loss_weight 0.3)."""
    x = avepool(net, f"{name}/ave_pool", bottom, 5, 3)
    x = conv(net, f"{name}/conv", x, 128, 1, std=0.08)
    x = fc(net, f"{name}/fc", x, 1024, std=0.02)
    layer(net, f"{name}/drop_fc", "Dropout", [x], [x],
          dropout_param=pb.DropoutParameter(dropout_ratio=0.7))
    lp = layer(net, f"{name}/classifier", "InnerProduct", [x],
               [f"{name}/classifier"],
               inner_product_param=pb.InnerProductParameter(
                   num_output=1000, weight_filler=_filler(type="xavier"),
                   bias_filler=_filler(type="constant")))
    lp.param = lr_params()
    ll = layer(net, f"{name}/loss", "SoftmaxWithLoss",
               [f"{name}/classifier", "label"], [f"{name}/loss"],
               phase=pb.Phase.TRAIN)
    ll.loss_weight = [0.3]


def googlenet(batch=128) -> pb.NetParameter:
    net = pb.NetParameter(name="GoogLeNet")
    memory_data(net, batch, 50, 3, 224, 224)
    x = conv(net, "conv1/7x7_s2", "data", 64, 7, stride=2, pad=3, std=0.03)
    x = maxpool(net, "pool1/3x3_s2", x, 3, 2)
    x = lrn(net, "pool1/norm1", x)
    x = conv(net, "conv2/3x3_reduce", x, 64, 1, std=0.09)
    x = conv(net, "conv2/3x3", x, 192, 3, pad=1, std=0.03)
    x = lrn(net, "conv2/norm2", x)
    x = maxpool(net, "pool2/3x3_s2", x, 3, 2)
    x = inception(net, "inception_3a", x, 64, 96, 128, 16, 32, 32)
    x = inception(net, "inception_3b", x, 128, 128, 192, 32, 96, 64)
    x = maxpool(net, "pool3/3x3_s2", x, 3, 2)
    x = inception(net, "inception_4a", x, 192, 96, 208, 16, 48, 64)
    _aux_head(net, "loss1", x)
    x = inception(net, "inception_4b", x, 160, 112, 224, 24, 64, 64)
    x = inception(net, "inception_4c", x, 128, 128, 256, 24, 64, 64)
    x = inception(net, "inception_4d", x, 112, 144, 288, 32, 64, 64)
    _aux_head(net, "loss2", x)
    x = inception(net, "inception_4e", x, 256, 160, 320, 32, 128, 128)
    x = maxpool(net, "pool4/3x3_s2", x, 3, 2)
    x = inception(net, "inception_5a", x, 256, 160, 320, 32, 128, 128)
    x = inception(net, "inception_5b", x, 384, 192, 384, 48, 128, 128)
    x = avepool(net, "pool5/7x7_s1", x, 7, 1)
    layer(net, "pool5/drop_7x7_s1", "Dropout", [x], [x],
          dropout_param=pb.DropoutParameter(dropout_ratio=0.4))
    lp = layer(net, "loss3/classifier", "InnerProduct", [x],
               ["loss3/classifier"],
               inner_product_param=pb.InnerProductParameter(
                   num_output=1000,
                   weight_filler=_filler(type="xavier"),
                   bias_filler=_filler(type="constant")))
    lp.param = lr_params()
    layer(net, "accuracy", "Accuracy", ["loss3/classifier", "label"],
          ["accuracy"], phase=pb.Phase.TEST)
    layer(net, "loss", "SoftmaxWithLoss", ["loss3/classifier", "label"],
          ["loss"])
    return net


def lenet_cos(batch=64) -> pb.NetParameter:
    """LeNet over a CoSData layer fed by DataFrameSource (reference
    lenet_cos_train_test.prototxt family)."""
    net = pb.NetParameter(name="LeNet")
    for phase, bs in ((pb.Phase.TRAIN, batch), (pb.Phase.TEST, 100)):
        lp = layer(net, "data", "CoSData", [], ["data", "label"],
                   phase=phase)
        lp.source_class = "com.yahoo.ml.caffe.DataFrameSource"
        cp = pb.CoSDataParameter(source="mnist_parquet", batch_size=bs)
        cp.top.append(pb.CoSTopParameter(
            name="data", type=pb.CoSTopType.RAW_IMAGE, channels=1,
            height=28, width=28,
            transform_param=pb.TransformationParameter(scale=0.00390625)))
        cp.top.append(pb.CoSTopParameter(name="label",
                                         type=pb.CoSTopType.INT))
        lp.cos_data_param = cp
    x = conv(net, "conv1", "data", 20, 5, relu=False)
    x = maxpool(net, "pool1", x, 2, 2)
    x = conv(net, "conv2", x, 50, 5, relu=False)
    x = maxpool(net, "pool2", x, 2, 2)
    x = fc(net, "ip1", x, 500, std=0.1, bias=0.0)
    x = fc(net, "ip2", x, 10, std=0.1, bias=0.0, relu=False)
    layer(net, "accuracy", "Accuracy", [x, "label"], ["accuracy"],
          phase=pb.Phase.TEST)
    layer(net, "loss", "SoftmaxWithLoss", [x, "label"], ["loss"])
    return net


def lrcn(batch=32, caption_len=21, vocab=8801) -> pb.NetParameter:
    """LRCN captioning net (factored 2-layer variant): CaffeNet conv stack
    + word Embed + 2x LSTM with the image feature as lstm2's static input,
    time-major captions — the reference's lrcn_cos.prototxt family."""
    net = pb.NetParameter(name="lrcn_caffenet_to_lstm")
    for phase in (pb.Phase.TRAIN, pb.Phase.TEST):
        lp = layer(net, "data", "CoSData", [],
                   ["data", "label", "cont_sentence", "input_sentence",
                    "target_sentence"], phase=phase)
        lp.source_class = "com.yahoo.ml.caffe.DataFrameSource"
        cp = pb.CoSDataParameter(source="coco_parquet", batch_size=batch)
        cp.top.append(pb.CoSTopParameter(
            name="data", type=pb.CoSTopType.ENCODED_IMAGE_WITH_DIM,
            channels=3, height=256, width=256, out_channels=3,
            out_height=227, out_width=227,
            transform_param=pb.TransformationParameter(
                mirror=True, crop_size=227,
                mean_value=[104.0, 117.0, 123.0])))
        cp.top.append(pb.CoSTopParameter(name="label",
                                         type=pb.CoSTopType.INT))
        for nm in ("cont_sentence", "input_sentence", "target_sentence"):
            cp.top.append(pb.CoSTopParameter(
                name=nm, type=pb.CoSTopType.INT_ARRAY, channels=caption_len,
                sample_num_axes=1, transpose=True))
        lp.cos_data_param = cp
    x = conv(net, "conv1", "data", 96, 11, stride=4, std=0.01, w_lr=0)
    x = maxpool(net, "pool1", x, 3, 2)
    x = lrn(net, "norm1", x)
    x = conv(net, "conv2", x, 256, 5, pad=2, group=2, bias=0.1, w_lr=0)
    x = maxpool(net, "pool2", x, 3, 2)
    x = lrn(net, "norm2", x)
    x = conv(net, "conv3", x, 384, 3, pad=1, w_lr=0)
    x = conv(net, "conv4", x, 384, 3, pad=1, group=2, bias=0.1, w_lr=0)
    x = conv(net, "conv5", x, 256, 3, pad=1, group=2, bias=0.1, w_lr=0)
    x = maxpool(net, "pool5", x, 3, 2)
    x = fc(net, "fc6", x, 4096, dropout=0.5)
    x = fc(net, "fc7", x, 4096, dropout=0.5)
    x = fc(net, "fc8", x, 1000, std=0.01, bias=0.0, relu=False)
    layer(net, "silence_label", "Silence", ["label"], [])
    emb = layer(net, "embedding", "Embed", ["input_sentence"],
                ["embedded_input_sentence"],
                embed_param=pb.EmbedParameter(
                    num_output=1000, input_dim=vocab, bias_term=False,
                    weight_filler=_filler(type="uniform", min=-0.08,
                                          max=0.08)))
    emb.param = [pb.ParamSpec(lr_mult=1.0)]
    rp = pb.RecurrentParameter(
        num_output=1000,
        weight_filler=_filler(type="uniform", min=-0.08, max=0.08),
        bias_filler=_filler(type="constant"))
    layer(net, "lstm1", "LSTM", ["embedded_input_sentence",
                                 "cont_sentence"], ["lstm1"],
          recurrent_param=rp)
    layer(net, "lstm2", "LSTM", ["lstm1", "cont_sentence", "fc8"],
          ["lstm2"], recurrent_param=rp.clone())
    pred = layer(net, "predict", "InnerProduct", ["lstm2"], ["predict"],
                 inner_product_param=pb.InnerProductParameter(
                     num_output=vocab, axis=2,
                     weight_filler=_filler(type="uniform", min=-0.08,
                                           max=0.08),
                     bias_filler=_filler(type="constant")))
    pred.param = lr_params()
    lp = layer(net, "cross_entropy_loss", "SoftmaxWithLoss",
               ["predict", "target_sentence"], ["cross_entropy_loss"],
               loss_param=pb.LossParameter(ignore_label=-1),
               softmax_param=pb.SoftmaxParameter(axis=2))
    lp.loss_weight = [20.0]
    layer(net, "accuracy", "Accuracy", ["predict", "target_sentence"],
          ["accuracy"], phase=pb.Phase.TEST,
          accuracy_param=pb.AccuracyParameter(axis=2, ignore_label=-1))
    return net


def solver(net_file, **kw) -> pb.SolverParameter:
    sp = pb.SolverParameter(net=net_file, **kw)
    return sp


def main():
    jobs = {
        "alexnet_train_test.prototxt": alexnet(),
        "cifar10_quick_train_test.prototxt": cifar10_quick(),
        "googlenet_train_test.prototxt": googlenet(),
        "lrcn_train_test.prototxt": lrcn(),
        "lenet_cos_train_test.prototxt": lenet_cos(),
    }
    for fname, net in jobs.items():
        with open(os.path.join(HERE, fname), "w") as fh:
            fh.write(f"# generated by caffeonspark_amd.models.generate\n")
            fh.write(text_format.dumps(net))
    solvers = {
        "alexnet_solver.prototxt": solver(
            "caffeonspark_amd/models/alexnet_train_test.prototxt",
            test_iter=[0], test_interval=0, base_lr=0.01, lr_policy="step",
            gamma=0.1, stepsize=100000, display=20, max_iter=450000,
            momentum=0.9, weight_decay=0.0005, snapshot=0,
            snapshot_prefix="alexnet", solver_mode=pb.SolverMode.GPU),
        "cifar10_quick_solver.prototxt": solver(
            "caffeonspark_amd/models/cifar10_quick_train_test.prototxt",
            test_iter=[100], test_interval=500, base_lr=0.001,
            lr_policy="fixed", display=100, max_iter=4000, momentum=0.9,
            weight_decay=0.004, snapshot=4000,
            snapshot_format=pb.SnapshotFormat.HDF5,
            snapshot_prefix="cifar10_quick", solver_mode=pb.SolverMode.GPU),
        "googlenet_solver.prototxt": solver(
            "caffeonspark_amd/models/googlenet_train_test.prototxt",
            test_iter=[0], test_interval=0, base_lr=0.01, lr_policy="poly",
            power=0.5, display=40, max_iter=2400000, momentum=0.9,
            weight_decay=0.0002, snapshot=0, snapshot_prefix="googlenet",
            solver_mode=pb.SolverMode.GPU),
        "lenet_cos_solver.prototxt": solver(
            "caffeonspark_amd/models/lenet_cos_train_test.prototxt",
            test_iter=[1], test_interval=500, base_lr=0.01, momentum=0.9,
            weight_decay=0.0005, lr_policy="inv", gamma=0.0001, power=0.75,
            display=100, max_iter=2000, snapshot=0,
            snapshot_prefix="lenet_cos", solver_mode=pb.SolverMode.GPU),
        "lrcn_solver.prototxt": solver(
            "caffeonspark_amd/models/lrcn_train_test.prototxt",
            test_iter=[0], test_interval=0, base_lr=0.01, lr_policy="step",
            gamma=0.5, stepsize=20000, display=1, max_iter=110000,
            momentum=0.9, weight_decay=0.0, snapshot=0,
            snapshot_prefix="lrcn", clip_gradients=10.0, average_loss=100,
            random_seed=1701, solver_mode=pb.SolverMode.GPU),
    }
    for fname, sp in solvers.items():
        with open(os.path.join(HERE, fname), "w") as fh:
            fh.write(f"# generated by caffeonspark_amd.models.generate\n")
            fh.write(text_format.dumps(sp))
    print("wrote", len(jobs) + len(solvers), "files to", HERE)


if __name__ == "__main__":
    main()
