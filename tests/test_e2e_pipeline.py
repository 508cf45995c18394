"""End-to-end pipeline: LMDB/SeqFile/DataFrame sources -> processor ->
solver -> test aggregation, mirroring the reference's own integration gates
(InterleaveTest accuracy>0.8, SourceTest round-trips — SURVEY.md §4)."""

import os

import numpy as np
import pytest
import torch

from caffeonspark_amd.api import CaffeOnSpark, Config
from caffeonspark_amd.data.lmdb_io import LmdbWriter
from caffeonspark_amd.data.processor import CaffeProcessor
from caffeonspark_amd.proto import caffe_pb, text_format
from caffeonspark_amd.tools.seq_value import datum_from_array

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def make_synthetic_lmdb(path, n, seed):
    """10-class separable 1x28x28 uint8 images: class-c has a bright
    vertical stripe at column 2c."""
    rng = np.random.RandomState(seed)
    items = []
    for i in range(n):
        label = rng.randint(0, 10)
        img = rng.randint(80, 150, size=(1, 28, 28)).astype(np.uint8)
        img[0, :, label * 2] = 250
        d = datum_from_array(img, label)
        items.append((f"{i:08d}".encode(), d.SerializeToString()))
    LmdbWriter(path).write(items)


LENET_NET = """
name: "LeNet"
layer {{
  name: "data" type: "MemoryData" top: "data" top: "label"
  include {{ phase: TRAIN }}
  source_class: "com.yahoo.ml.caffe.LMDB"
  memory_data_param {{ source: "{train}" batch_size: 64
                       channels: 1 height: 28 width: 28 }}
  transform_param {{ scale: 0.00390625 mean_value: 128 }}
}}
layer {{
  name: "data" type: "MemoryData" top: "data" top: "label"
  include {{ phase: TEST }}
  source_class: "com.yahoo.ml.caffe.LMDB"
  memory_data_param {{ source: "{test}" batch_size: 100
                       channels: 1 height: 28 width: 28 }}
  transform_param {{ scale: 0.00390625 mean_value: 128 }}
}}
layer {{ name: "conv1" type: "Convolution" bottom: "data" top: "conv1"
  param {{ lr_mult: 1 }} param {{ lr_mult: 2 }}
  convolution_param {{ num_output: 20 kernel_size: 5
    weight_filler {{ type: "xavier" }} }} }}
layer {{ name: "pool1" type: "Pooling" bottom: "conv1" top: "pool1"
  pooling_param {{ pool: MAX kernel_size: 2 stride: 2 }} }}
layer {{ name: "ip1" type: "InnerProduct" bottom: "pool1" top: "ip1"
  param {{ lr_mult: 1 }} param {{ lr_mult: 2 }}
  inner_product_param {{ num_output: 100
    weight_filler {{ type: "xavier" }} }} }}
layer {{ name: "relu1" type: "ReLU" bottom: "ip1" top: "ip1" }}
layer {{ name: "ip2" type: "InnerProduct" bottom: "ip1" top: "ip2"
  param {{ lr_mult: 1 }} param {{ lr_mult: 2 }}
  inner_product_param {{ num_output: 10
    weight_filler {{ type: "xavier" }} }} }}
layer {{ name: "accuracy" type: "Accuracy" bottom: "ip2" bottom: "label"
  top: "accuracy" include {{ phase: TEST }} }}
layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "ip2" bottom: "label"
  top: "loss" }}
"""

SOLVER = """
net: "{net}"
test_iter: 2
test_interval: {test_interval}
base_lr: 0.01
momentum: 0.9
weight_decay: 0.0005
lr_policy: "fixed"
display: 0
max_iter: {max_iter}
snapshot: 0
snapshot_prefix: "{prefix}"
random_seed: 11
"""


@pytest.fixture(scope="module")
def workdir(tmp_path_factory):
    d = tmp_path_factory.mktemp("e2e")
    make_synthetic_lmdb(str(d / "train_lmdb"), 1200, seed=1)
    make_synthetic_lmdb(str(d / "test_lmdb"), 200, seed=2)
    net_file = d / "lenet.prototxt"
    net_file.write_text(LENET_NET.format(train=str(d / "train_lmdb"),
                                         test=str(d / "test_lmdb")))
    return d


def _solver_file(workdir, name, test_interval=0, max_iter=120):
    f = workdir / name
    f.write_text(SOLVER.format(net=str(workdir / "lenet.prototxt"),
                               test_interval=test_interval,
                               max_iter=max_iter,
                               prefix=str(workdir / "lenet")))
    return str(f)


def test_train_and_test_via_facade(workdir):
    """BASELINE config 1: LeNet via memory source, CPU — the reference's
    plumbing-correctness gate (accuracy > 0.8)."""
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    conf = Config(["-conf", _solver_file(workdir, "solver.prototxt"),
                   "-train", "-label", "label"])
    cos = CaffeOnSpark(conf)
    cos.train()
    # snapshot written by train (snapshot_after_train)
    snaps = [f for f in os.listdir(workdir) if f.endswith(".caffemodel")]
    assert snaps, "no snapshot written"

    CaffeProcessor.reset_instance()
    conf2 = Config(["-conf", _solver_file(workdir, "solver2.prototxt"),
                    "-test", "-weights",
                    str(workdir / sorted(snaps)[-1])])
    cos2 = CaffeOnSpark(conf2)
    result = cos2.test(max_samples=200)
    assert "accuracy" in result
    acc = result["accuracy"][0]
    assert acc > 0.8, f"accuracy {acc}"
    CaffeProcessor.reset_instance()


def test_train_with_validation(workdir):
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    conf = Config(["-conf",
                   _solver_file(workdir, "solver3.prototxt",
                                test_interval=50, max_iter=110),
                   "-train"])
    cos = CaffeOnSpark(conf)
    results = cos.train_with_validation()
    assert len(results) >= 1
    assert "accuracy" in results[-1]
    CaffeProcessor.reset_instance()


def test_features_extraction(workdir):
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    out = str(workdir / "features.json")
    conf = Config(["-conf", _solver_file(workdir, "solver4.prototxt"),
                   "-features", "ip2", "-label", "label",
                   "-output", out, "-outputFormat", "json"])
    cos = CaffeOnSpark(conf)
    df = cos.features(max_samples=100)
    assert "SampleID" in df.columns and "ip2" in df.columns
    assert len(df) >= 100
    assert len(df["ip2"][0]) == 10
    assert os.path.exists(out)
    CaffeProcessor.reset_instance()

    # parquet output format (reference -outputFormat parquet path)
    pytest.importorskip("pyarrow")
    out_pq = str(workdir / "features.parquet")
    conf2 = Config(["-conf", _solver_file(workdir, "solver4b.prototxt"),
                    "-features", "ip2", "-label", "label",
                    "-output", out_pq, "-outputFormat", "parquet"])
    cos2 = CaffeOnSpark(conf2)
    cos2.features(max_samples=20)
    import pandas as pd
    back = pd.read_parquet(out_pq)
    assert "ip2" in back.columns and len(back) >= 20
    CaffeProcessor.reset_instance()


def test_seqfile_source_roundtrip(workdir, tmp_path):
    """SourceTest analog: LMDB -> SequenceFile -> read back."""
    from caffeonspark_amd.data.seqfile import SequenceFileReader
    from caffeonspark_amd.tools.converters import lmdb2sequence

    seq = str(tmp_path / "train.seq")
    n = lmdb2sequence(str(workdir / "train_lmdb"), seq)
    assert n == 1200
    r = SequenceFileReader(seq)
    items = list(r.items())
    assert len(items) == 1200
    d = caffe_pb.Datum.FromString(items[0][1])
    assert (d.channels, d.height, d.width) == (1, 28, 28)


def test_dataframe_source(workdir, tmp_path):
    """LMDB -> parquet -> ImageDataFrame source feeds a batch."""
    from caffeonspark_amd.data.source import get_source
    from caffeonspark_amd.tools.converters import lmdb2dataframe

    pq_file = str(tmp_path / "train.parquet")
    lmdb2dataframe(str(workdir / "train_lmdb"), pq_file)

    net = text_format.parse_file(str(workdir / "lenet.prototxt"),
                                 caffe_pb.NetParameter)
    lp = net.layer[0]
    lp.source_class = "com.yahoo.ml.caffe.ImageDataFrame"
    lp.memory_data_param.source = pq_file

    class FakeConf:
        net_param = net
        seed = 1

    src = get_source(FakeConf(), True)
    src.init()
    it = src.sample_iter(0, 1, epochs=1)
    for _ in range(70):
        src.offer(next(it))
    batch = src.next_batch(torch.device("cpu"), torch.float32)
    assert batch[0].shape == (64, 1, 28, 28)
    assert batch[1].shape == (64,)


def test_cos_dataframe_training(workdir, tmp_path):
    """lenet_cos: CoSData layer fed by DataFrameSource (parquet), the
    reference's lenet_cos_train_test.prototxt path."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from caffeonspark_amd.proto import text_format

    rng = np.random.RandomState(7)
    ids, datas, labels = [], [], []
    for i in range(800):
        label = rng.randint(0, 10)
        img = rng.randint(80, 150, size=(1, 28, 28)).astype(np.uint8)
        img[0, :, label * 2] = 250
        ids.append(str(i))
        datas.append(img.tobytes())
        labels.append(label)
    pq_file = str(tmp_path / "mnist.parquet")
    pq.write_table(pa.table({"id": ids, "data": datas, "label": labels}),
                   pq_file)

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    net = text_format.parse_file(
        os.path.join(root, "caffeonspark_amd", "models",
                     "lenet_cos_train_test.prototxt"), caffe_pb.NetParameter)
    for lp in net.layer:
        if lp.type == "CoSData":
            lp.cos_data_param.source = pq_file
    net_file = tmp_path / "lenet_cos.prototxt"
    net_file.write_text(text_format.dumps(net))
    solver_file = tmp_path / "solver_cos.prototxt"
    solver_file.write_text(f"""
net: "{net_file}"
test_iter: 0
test_interval: 0
base_lr: 0.01
momentum: 0.9
lr_policy: "fixed"
display: 0
max_iter: 60
snapshot: 0
snapshot_prefix: "{tmp_path}/lenet_cos"
random_seed: 3
""")
    CaffeProcessor.reset_instance()
    os.chdir(tmp_path)
    conf = Config(["-conf", str(solver_file), "-train"])
    cos = CaffeOnSpark(conf)
    cos.train()
    # training ran to max_iter and snapshotted
    snaps = [f for f in os.listdir(tmp_path) if f.endswith(".caffemodel")]
    assert snaps
    CaffeProcessor.reset_instance()


def test_ml_pipeline_example(workdir):
    """MyMLPipeline analog: DL features -> scikit-learn classifier."""
    pytest.importorskip("sklearn")
    import sys
    sys.path.insert(0, os.path.join(ROOT, "examples"))
    import ml_pipeline

    CaffeProcessor.reset_instance()
    snaps = [f for f in os.listdir(workdir) if f.endswith(".caffemodel")]
    if not snaps:
        pytest.skip("depends on test_train_and_test_via_facade snapshot")
    ml_pipeline.main(["-conf", str(workdir / "solver.prototxt"),
                      "-weights", str(workdir / sorted(snaps)[-1]),
                      "-features", "ip1", "-label", "label"])
    CaffeProcessor.reset_instance()


def test_image_caption_example(tmp_path):
    """ImageCaption.py analog: greedy LRCN decode over our Net (random
    weights; checks the decode loop plumbing, not caption quality)."""
    import sys

    import numpy as np
    sys.path.insert(0, os.path.join(ROOT, "examples"))
    import image_caption

    from caffeonspark_amd.api import Config

    try:
        from PIL import Image
    except ImportError:
        pytest.skip("PIL not available")
    img = (np.random.RandomState(0).rand(64, 64, 3) * 255).astype("uint8")
    p = str(tmp_path / "cat.jpg")
    Image.fromarray(img).save(p)

    solver = os.path.join(ROOT, "caffeonspark_amd", "models",
                          "lrcn_solver.prototxt")
    conf = Config(["-conf", solver])
    results = image_caption.caption_images(conf, [p], max_len=2)
    assert len(results) == 1
    path, caption = results[0]
    assert path == p
    assert isinstance(caption, str)


@pytest.mark.gpu
def test_gpu_end_to_end_training(tmp_path):
    """Full pipeline on the GPU bf16 path: LMDB source -> facade train to
    convergence -> binaryproto snapshot -> restore into a fresh solver ->
    loss parity.  Proves the HIP kernel path trains, not just matches
    single-op numerics."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    make_synthetic_lmdb(str(tmp_path / "train_lmdb"), 1200, seed=1)
    net_file = tmp_path / "lenet.prototxt"
    net_file.write_text(LENET_NET.format(train=str(tmp_path / "train_lmdb"),
                                         test=str(tmp_path / "train_lmdb")))
    f = tmp_path / "solver.prototxt"
    f.write_text(SOLVER.format(net=str(net_file), test_interval=0,
                               max_iter=150, prefix=str(tmp_path / "g")))
    CaffeProcessor.reset_instance()
    os.chdir(tmp_path)
    conf = Config(["-conf", str(f), "-train", "-label", "label",
                   "-dtype", "bf16"])
    assert conf.device.type == "cuda"
    cos = CaffeOnSpark(conf)
    cos.train()
    snaps = sorted(fn for fn in os.listdir(tmp_path)
                   if fn.endswith(".caffemodel"))
    states = sorted(fn for fn in os.listdir(tmp_path)
                    if fn.endswith(".solverstate"))
    assert snaps and states
    CaffeProcessor.reset_instance()

    # restore on GPU and check training state round-trips
    from caffeonspark_amd.core import solver_from_prototxt
    s2 = solver_from_prototxt(str(f), device=torch.device("cuda", 0),
                              dtype=torch.bfloat16)
    s2.restore(str(tmp_path / states[-1]))
    assert s2.iter == 150
    # evaluate on a batch drawn from the training distribution (stripe at
    # column 2*label), with the net's transform (x-128)/256 applied
    rng = np.random.RandomState(9)
    imgs, labels = [], []
    for _ in range(64):
        label = rng.randint(0, 10)
        img = rng.randint(80, 150, size=(1, 28, 28)).astype(np.float32)
        img[0, :, label * 2] = 250
        imgs.append((img - 128.0) * 0.00390625)
        labels.append(label)
    x = torch.tensor(np.stack(imgs)).to("cuda", torch.bfloat16)
    y = torch.tensor(labels, dtype=torch.float32, device="cuda")
    s2.net.data_layers()[0].reset(x, y)
    loss = s2.net.forward()
    assert loss < 1.0, f"restored net loss {loss} (random init would be ~2.3)"


def test_vector_mean_and_config_parse():
    """VectorMean UDAF analog + Config flag-surface parsing."""
    import pandas as pd

    from caffeonspark_amd.api.vector_mean import vector_mean
    col = pd.Series([[1.0, 2.0], [3.0, 4.0], [5.0, 6.0]])
    assert vector_mean(col) == [3.0, 4.0]

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    solver = os.path.join(root, "caffeonspark_amd", "models",
                          "lenet_memory_solver.prototxt")
    conf = Config(["-conf", solver, "-train", "-devices", "4",
                   "-clusterSize", "2", "-connection", "ethernet",
                   "-outputFormat", "parquet", "-lmdb_partitions", "7",
                   "-captionLength", "15", "-vocabSize", "999"])
    assert conf.devices == 4
    assert conf.clusterSize == 2
    assert conf.outputFormat == "parquet"
    assert conf.lmdbPartitions == 7
    assert conf.captionLength == 15
    assert conf.vocabSize == 999
    assert conf.isTraining


def test_cli_main_train_and_test(workdir, capsys):
    """The reference CLI entry (CaffeOnSpark.main): -train then -test
    through the module main()."""
    from caffeonspark_amd.api import caffe_on_spark
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    caffe_on_spark.main(["-conf",
                         _solver_file(workdir, "solver_cli.prototxt",
                                      max_iter=60),
                         "-train", "-label", "label"])
    snaps = sorted(f for f in os.listdir(workdir)
                   if f.endswith(".caffemodel"))
    assert snaps
    CaffeProcessor.reset_instance()
    caffe_on_spark.main(["-conf",
                         _solver_file(workdir, "solver_cli2.prototxt"),
                         "-test", "-weights", str(workdir / snaps[-1])])
    out = capsys.readouterr().out
    assert "accuracy" in out
    CaffeProcessor.reset_instance()


def test_multiclass_lr_example(workdir):
    """MultiClassLogisticRegression.py analog end-to-end."""
    pytest.importorskip("sklearn")
    import sys
    sys.path.insert(0, os.path.join(ROOT, "examples"))
    import multiclass_logistic_regression as mlr

    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    snaps = sorted(f for f in os.listdir(workdir)
                   if f.endswith(".caffemodel"))
    if not snaps:
        pytest.skip("depends on an earlier training snapshot")
    acc = mlr.main(["-conf", str(workdir / "solver.prototxt"),
                    "-weights", str(workdir / snaps[-1]),
                    "-features", "ip1", "-label", "label"])
    assert acc > 0.5
    CaffeProcessor.reset_instance()


def test_model_flag_copies_snapshot(workdir):
    """-model file:PATH: the final model is copied there (reference
    FSUtils.GenModelOrState)."""
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    dest = str(workdir / "final_model.caffemodel")
    conf = Config(["-conf", _solver_file(workdir, "solver_m.prototxt",
                                         max_iter=40),
                   "-train", "-label", "label",
                   "-model", "file:" + dest])
    CaffeOnSpark(conf).train()
    assert os.path.exists(dest)
    CaffeProcessor.reset_instance()


def test_features_multi_blob(workdir):
    """-features a,b: one row carries every requested blob."""
    CaffeProcessor.reset_instance()
    os.chdir(workdir)
    conf = Config(["-conf", _solver_file(workdir, "solver_mb.prototxt"),
                   "-features", "ip1,ip2", "-label", "label"])
    cos = CaffeOnSpark(conf)
    df = cos.features(max_samples=50)
    assert {"SampleID", "label", "ip1", "ip2"} <= set(df.columns)
    assert len(df["ip1"][0]) == 100 and len(df["ip2"][0]) == 10
    CaffeProcessor.reset_instance()


def test_notebooks_valid():
    """Reference ships example notebooks; ours must be valid nbformat-4
    JSON whose code cells at least compile."""
    import json

    nbdir = os.path.join(ROOT, "examples", "notebooks")
    nbs = [f for f in os.listdir(nbdir) if f.endswith(".ipynb")]
    assert len(nbs) >= 2
    for name in nbs:
        with open(os.path.join(nbdir, name)) as f:
            nb = json.load(f)
        assert nb["nbformat"] == 4
        for cell in nb["cells"]:
            if cell["cell_type"] == "code":
                compile("".join(cell["source"]), name, "exec")
