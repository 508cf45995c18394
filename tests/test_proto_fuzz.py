"""Property-based round-trip tests of the clean-room proto codec — the
analog of the reference's randomized Python-bridge conversion tests
(TestResources.py).  hypothesis drives random field values through text
and wire round-trips."""

import math

from hypothesis import given, settings
from hypothesis import strategies as st

from caffeonspark_amd.proto import caffe_pb, text_format

f32 = st.floats(min_value=-3.0000000054977558e+38, max_value=3.0000000054977558e+38, width=32)
i32 = st.integers(min_value=-(2 ** 31), max_value=2 ** 31 - 1)
u32 = st.integers(min_value=0, max_value=2 ** 32 - 1)
name = st.text(
    st.characters(whitelist_categories=("L", "N"), max_codepoint=0x2FF),
    max_size=24)


@settings(max_examples=200, deadline=None)
@given(base_lr=f32, max_iter=u32, gamma=f32, momentum=f32,
       snapshot_prefix=name, stepvalue=st.lists(u32, max_size=6),
       test_iter=st.lists(u32, max_size=4), iter_size=i32)
def test_solver_param_roundtrips(base_lr, max_iter, gamma, momentum,
                                 snapshot_prefix, stepvalue, test_iter,
                                 iter_size):
    p = caffe_pb.SolverParameter(
        base_lr=base_lr, max_iter=max_iter, gamma=gamma,
        momentum=momentum, snapshot_prefix=snapshot_prefix,
        stepvalue=stepvalue, test_iter=test_iter, iter_size=iter_size)
    # binary wire
    b = caffe_pb.SolverParameter.FromString(p.SerializeToString())
    # text format
    t = text_format.parse(text_format.dumps(p), caffe_pb.SolverParameter)
    for q in (b, t):
        assert math.isclose(q.base_lr, base_lr, rel_tol=1e-6, abs_tol=1e-37)
        assert q.max_iter == max_iter
        assert q.snapshot_prefix == snapshot_prefix
        assert list(q.stepvalue) == stepvalue
        assert list(q.test_iter) == test_iter
        assert q.iter_size == iter_size


@settings(max_examples=100, deadline=None)
@given(label=i32, channels=u32, data=st.binary(max_size=512),
       floats=st.lists(f32, max_size=32))
def test_datum_roundtrips(label, channels, data, floats):
    d = caffe_pb.Datum(label=label, channels=channels, data=data,
                       float_data=floats)
    b = caffe_pb.Datum.FromString(d.SerializeToString())
    assert b.label == label and b.channels == channels
    assert bytes(b.data) == data
    assert len(b.float_data) == len(floats)
    for a, e in zip(b.float_data, floats):
        assert math.isclose(a, e, rel_tol=1e-6, abs_tol=1e-37)


@settings(max_examples=100, deadline=None)
@given(nm=name, lr=st.lists(f32, max_size=3), bottoms=st.lists(name,
                                                               max_size=4))
def test_layer_param_roundtrips(nm, lr, bottoms):
    lp = caffe_pb.LayerParameter(name=nm, type="ReLU", bottom=bottoms,
                                 loss_weight=lr)
    np_ = caffe_pb.NetParameter(name=nm, layer=[lp])
    b = caffe_pb.NetParameter.FromString(np_.SerializeToString())
    t = text_format.parse(text_format.dumps(np_), caffe_pb.NetParameter)
    for q in (b, t):
        assert q.name == nm
        assert q.layer[0].name == nm
        assert list(q.layer[0].bottom) == bottoms


# --------------------------------------------------------- format fuzz

kv_key = st.binary(min_size=1, max_size=48)
kv_val = st.binary(max_size=3000)


@settings(max_examples=40, deadline=None)
@given(items=st.dictionaries(kv_key, kv_val, max_size=120))
def test_lmdb_roundtrip_fuzz(items, tmp_path_factory):
    """Random key/value sets (sizes spanning inline and overflow pages)
    survive the clean-room LMDB writer -> reader round-trip in key
    order."""
    from caffeonspark_amd.data.lmdb_io import LmdbReader, LmdbWriter
    d = tmp_path_factory.mktemp("lmdbfuzz")
    path = str(d / "db")
    pairs = sorted(items.items())
    LmdbWriter(path).write(list(items.items()))
    got = list(LmdbReader(path).items())
    assert got == pairs


@settings(max_examples=40, deadline=None)
@given(items=st.lists(st.tuples(st.binary(min_size=1, max_size=64),
                                st.binary(max_size=2000)),
                      max_size=60))
def test_seqfile_roundtrip_fuzz(items, tmp_path_factory):
    """Random records round-trip through the Hadoop SequenceFile codec
    (sync markers land wherever they land)."""
    from caffeonspark_amd.data.seqfile import SequenceFileReader, \
        SequenceFileWriter
    d = tmp_path_factory.mktemp("seqfuzz")
    p = str(d / "f.seq")
    with SequenceFileWriter(p) as w:
        for k, v in items:
            w.append(k, v)
    assert list(SequenceFileReader(p).items()) == items


@settings(max_examples=30, deadline=None)
@given(arrays=st.dictionaries(
    st.text(st.characters(whitelist_categories=("L", "N")), min_size=1,
            max_size=12),
    st.lists(f32,
             min_size=1, max_size=64),
    min_size=1, max_size=8),
    attr=st.text(max_size=40))
def test_hdf5_roundtrip_fuzz(arrays, attr, tmp_path_factory):
    """Random group trees of float arrays + a string attribute survive
    the minimal HDF5 v0 writer -> reader round-trip."""
    import numpy as np

    from caffeonspark_amd.utils.hdf5 import H5Group, load, save
    d = tmp_path_factory.mktemp("h5fuzz")
    p = str(d / "t.h5")
    root = H5Group()
    root.attrs["note"] = attr
    sub = H5Group()
    for name, vals in arrays.items():
        sub[name] = np.asarray(vals, dtype=np.float32)
    root["data"] = sub
    save(p, root)
    back = load(p)
    assert back.attrs.get("note", "") == attr
    got = back["data"]
    assert set(got.keys()) == set(arrays.keys())
    for name, vals in arrays.items():
        np.testing.assert_allclose(
            np.asarray(got[name], dtype=np.float32),
            np.asarray(vals, dtype=np.float32), rtol=1e-6)


@settings(max_examples=150, deadline=None)
@given(data=st.lists(f32, max_size=24), num=i32, s=st.text(max_size=32),
       u=st.integers(min_value=0, max_value=2 ** 64 - 1),
       flag=st.booleans())
def test_wire_fuzz_against_google_protobuf(data, num, s, u, flag):
    """Random values encoded by google.protobuf decode identically with
    our codec, and vice versa (both directions, 5 field kinds)."""
    import math as _math

    import pytest as _pytest
    _pytest.importorskip("google.protobuf")
    from google.protobuf import descriptor_pb2, descriptor_pool, \
        message_factory

    from caffeonspark_amd.proto.pbcodec import Field, Message

    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "fz.proto"
    fdp.package = "fz"
    m = fdp.message_type.add()
    m.name = "M"
    for fname, num_, typ, label, packed in (
            ("data", 5, 2, 3, True),     # repeated packed float
            ("num", 1, 5, 1, False),     # int32
            ("s", 20, 9, 1, False),      # string
            ("u", 7, 4, 1, False),       # uint64
            ("flag", 9, 8, 1, False)):   # bool
        f = m.field.add()
        f.name, f.number, f.type, f.label = fname, num_, typ, label
        if packed:
            f.options.packed = True
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("fz.M"))

    class M(Message):
        FIELDS = [
            Field(5, "data", "float", repeated=True, packed=True),
            Field(1, "num", "int32"),
            Field(20, "s", "string"),
            Field(7, "u", "uint64"),
            Field(9, "flag", "bool"),
        ]

    g = cls()
    g.data.extend(data)
    g.num = num
    g.s = s
    g.u = u
    g.flag = flag
    ours = M.FromString(g.SerializeToString())
    assert ours.num == num and ours.s == s and ours.u == u
    assert bool(ours.flag) == flag
    assert len(ours.data) == len(data)
    for a, e in zip(ours.data, data):
        assert _math.isclose(a, e, rel_tol=1e-6, abs_tol=1e-37)

    g2 = cls()
    g2.ParseFromString(ours.SerializeToString())
    assert g2.num == num and g2.s == s and g2.u == u
    assert bool(g2.flag) == flag
    assert len(g2.data) == len(data)
