"""Proto codec tests: text-format parse, binary wire round-trip, and
cross-validation of the wire format against google.protobuf."""

import pytest

from caffeonspark_amd.proto import caffe_pb, text_format

LENET = "caffeonspark_amd/models/lenet_memory_train_test.prototxt"


def test_parse_lenet():
    net = text_format.parse_file(LENET, caffe_pb.NetParameter)
    assert net.name == "LeNet"
    assert len(net.layer) == 11
    types = [l.type for l in net.layer]
    assert types.count("MemoryData") == 2
    conv1 = [l for l in net.layer if l.name == "conv1"][0]
    assert conv1.convolution_param.num_output == 20
    assert list(conv1.convolution_param.kernel_size) == [5]
    assert conv1.param[0].lr_mult == 1.0
    assert conv1.param[1].lr_mult == 2.0
    assert conv1.convolution_param.weight_filler.type == "xavier"


def test_parse_defaults():
    lp = text_format.parse(
        'name: "x" type: "Pooling" pooling_param { pool: MAX kernel_size: 2 }',
        caffe_pb.LayerParameter)
    assert lp.pooling_param.pool == caffe_pb.PoolingParameter.PoolMethod.MAX
    assert lp.pooling_param.stride == 1  # default
    # unset solver defaults
    sp = caffe_pb.SolverParameter()
    assert sp.momentum2 == 0.999
    assert sp.iter_size == 1
    assert sp.type == "SGD"


def test_single_quotes_and_comments():
    sp = text_format.parse(
        "# a comment\ntrain_state: { stage: 'factored' stage: '2-layer' }\n"
        "base_lr: 0.01 # trailing\n", caffe_pb.SolverParameter)
    assert list(sp.train_state.stage) == ["factored", "2-layer"]
    assert sp.base_lr == pytest.approx(0.01)


def test_text_round_trip():
    net = text_format.parse_file(LENET, caffe_pb.NetParameter)
    text = text_format.dumps(net)
    net2 = text_format.parse(text, caffe_pb.NetParameter)
    assert net == net2


def test_binary_round_trip():
    net = text_format.parse_file(LENET, caffe_pb.NetParameter)
    data = net.SerializeToString()
    net2 = caffe_pb.NetParameter.FromString(data)
    assert net == net2


def test_packed_blob():
    bp = caffe_pb.BlobProto()
    bp.shape = caffe_pb.BlobShape(dim=[2, 3])
    bp.data = [1.0, 2.0, 3.0, 4.0, 5.0, 6.0]
    enc = bp.SerializeToString()
    bp2 = caffe_pb.BlobProto.FromString(enc)
    assert list(bp2.shape.dim) == [2, 3]
    assert list(bp2.data) == [1.0, 2.0, 3.0, 4.0, 5.0, 6.0]


def test_wire_compat_with_google_protobuf():
    """Encode with google.protobuf (dynamic message mirroring BlobProto's
    field layout) and decode with our codec — asserts wire compatibility."""
    pb2 = pytest.importorskip("google.protobuf")
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "x.proto"
    fdp.package = "x"
    m = fdp.message_type.add()
    m.name = "B"
    f = m.field.add()
    f.name, f.number, f.type, f.label = "data", 5, 2, 3  # float, repeated
    f.options.packed = True
    f = m.field.add()
    f.name, f.number, f.type, f.label = "num", 1, 5, 1  # int32, optional
    f = m.field.add()
    f.name, f.number, f.type, f.label = "s", 20, 9, 1  # string, optional
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    desc = pool.FindMessageTypeByName("x.B")
    cls = message_factory.GetMessageClass(desc)
    gm = cls()
    gm.data.extend([1.5, -2.5, 3.25])
    gm.num = -7
    gm.s = "hello"
    wire = gm.SerializeToString()

    from caffeonspark_amd.proto.pbcodec import Field, Message

    class B(Message):
        FIELDS = [
            Field(5, "data", "float", repeated=True, packed=True),
            Field(1, "num", "int32"),
            Field(20, "s", "string"),
        ]

    ours = B.FromString(wire)
    assert list(ours.data) == [1.5, -2.5, 3.25]
    assert ours.num == -7
    assert ours.s == "hello"
    # and the reverse direction
    wire2 = ours.SerializeToString()
    gm2 = cls()
    gm2.ParseFromString(wire2)
    assert list(gm2.data) == [1.5, -2.5, 3.25]
    assert gm2.num == -7
    assert gm2.s == "hello"


def test_negative_int32_varint():
    d = caffe_pb.Datum(label=-1)
    d2 = caffe_pb.Datum.FromString(d.SerializeToString())
    assert d2.label == -1


def test_unknown_field_skipped():
    # encode a message with a field our schema doesn't know (e.g. 999)
    from caffeonspark_amd.proto.pbcodec import Field, Message

    class Ext(Message):
        FIELDS = [Field(999, "mystery", "string"), Field(1, "num", "int32")]

    wire = Ext(mystery="zzz", num=3).SerializeToString()
    d = caffe_pb.Datum.FromString(wire)  # Datum has field 1 (channels)
    assert d.channels == 3


def test_merge_semantics():
    a = caffe_pb.NetParameter(name="a")
    a.layer.append(caffe_pb.LayerParameter(name="l1", type="ReLU"))
    b = caffe_pb.NetParameter()
    b.CopyFrom(a)
    b.layer[0].name = "changed"
    assert a.layer[0].name == "l1"  # deep copy


def test_text_format_string_escapes():
    """Escaped strings round-trip through the text codec (caffe prototxts
    in the wild use quotes and backslashes in paths)."""
    from caffeonspark_amd.proto import caffe_pb, text_format
    p = caffe_pb.NetParameter(name='weird "name"\\with\tescapes\n')
    txt = text_format.dumps(p)
    back = text_format.parse(txt, caffe_pb.NetParameter)
    assert back.name == p.name


def test_bytes_field_roundtrip():
    """Datum.data (bytes) with all byte values survives text + binary."""
    from caffeonspark_amd.proto import caffe_pb
    raw = bytes(range(256))
    d = caffe_pb.Datum(channels=1, height=16, width=16, label=3, data=raw)
    back = caffe_pb.Datum.FromString(d.SerializeToString())
    assert bytes(back.data) == raw and back.label == 3
