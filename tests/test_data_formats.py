"""Format-level unit tests: LMDB edge cases, SequenceFile sync markers,
DataTransformer semantics."""

import numpy as np
import pytest
import torch

from caffeonspark_amd.data.lmdb_io import LmdbReader, LmdbWriter
from caffeonspark_amd.data.seqfile import SequenceFileReader, \
    SequenceFileWriter
from caffeonspark_amd.data.transformer import DataTransformer
from caffeonspark_amd.proto import caffe_pb


def test_lmdb_empty(tmp_path):
    p = str(tmp_path / "empty")
    LmdbWriter(p).write([])
    r = LmdbReader(p)
    assert len(list(r.items())) == 0
    assert len(r) == 0


def test_lmdb_single_and_sorted(tmp_path):
    p = str(tmp_path / "one")
    items = [(b"zz", b"last"), (b"aa", b"first")]
    LmdbWriter(p).write(items)
    got = list(LmdbReader(p).items())
    assert got == sorted(items)  # B+tree iterates in key order


def test_lmdb_large_values_multilevel(tmp_path):
    """Overflow pages + multi-level branch tree."""
    rng = np.random.RandomState(3)
    items = [(f"k{i:06d}".encode(), rng.bytes(6000)) for i in range(2000)]
    p = str(tmp_path / "big")
    LmdbWriter(p).write(items)
    got = list(LmdbReader(p).items())
    assert len(got) == 2000
    assert got[1234] == sorted(items)[1234]


def test_seqfile_sync_markers(tmp_path):
    """Records spanning many sync intervals round-trip."""
    p = str(tmp_path / "s.seq")
    items = [(f"key{i}".encode(), bytes([i % 251]) * 300)
             for i in range(100)]
    with SequenceFileWriter(p) as w:
        for k, v in items:
            w.append(k, v)
    r = SequenceFileReader(p)
    assert r.key_class == "org.apache.hadoop.io.BytesWritable"
    assert list(r.items()) == items


def test_transformer_crop_and_mean():
    tp = caffe_pb.TransformationParameter(
        crop_size=4, mean_value=[10.0], scale=0.5)
    xf = DataTransformer(tp, caffe_pb.Phase.TEST)  # center crop, no mirror
    img = np.arange(36, dtype=np.uint8).reshape(6, 6, 1)
    out = xf.transform_one(img)
    assert out.shape == (1, 4, 4)
    # center crop offset (6-4)//2 = 1; value = (pixel - 10) * 0.5
    assert out[0, 0, 0] == pytest.approx((img[1, 1, 0] - 10.0) * 0.5)


def test_transformer_mirror_deterministic_seed():
    tp = caffe_pb.TransformationParameter(mirror=True)
    a = DataTransformer(tp, caffe_pb.Phase.TRAIN, seed=3)
    b = DataTransformer(tp, caffe_pb.Phase.TRAIN, seed=3)
    img = np.random.RandomState(0).randint(0, 255, (5, 5, 3)).astype(
        np.uint8)
    for _ in range(5):
        np.testing.assert_array_equal(a.transform_one(img),
                                      b.transform_one(img))


def test_transformer_batch_shape():
    tp = caffe_pb.TransformationParameter()
    xf = DataTransformer(tp, caffe_pb.Phase.TEST)
    imgs = [np.zeros((8, 8, 3), dtype=np.uint8) for _ in range(4)]
    out = xf.transform(imgs)
    assert isinstance(out, torch.Tensor)
    assert out.shape == (4, 3, 8, 8)


def test_lmdb_source_rank_sharding(tmp_path):
    """sample_iter(rank, world): ranks see disjoint round-robin shards
    whose union is the full dataset (reference: LmdbRDD partitioning)."""
    from caffeonspark_amd.data.lmdb_source import LMDBSource
    from caffeonspark_amd.proto import caffe_pb as pb

    path = str(tmp_path / "db")
    items = []
    for i in range(17):
        d = pb.Datum(channels=1, height=2, width=2, label=i % 5,
                     data=bytes([i] * 4))
        items.append((f"{i:04d}".encode(), d.SerializeToString()))
    LmdbWriter(path).write(items)

    class _Conf:
        lmdb_partitions = 0

    class _LP:
        class memory_data_param:
            source = path
            batch_size = 4
            channels, height, width = 1, 2, 2

        def __getattr__(self, k):
            raise AttributeError(k)

    import types
    src = LMDBSource.__new__(LMDBSource)
    src.conf = types.SimpleNamespace(isRddPersistent=False)
    from caffeonspark_amd.data.lmdb_io import LmdbReader
    src.reader = LmdbReader(path)
    shards = [[s.id for s in src.sample_iter(rank=r, world=3, epochs=1)]
              for r in range(3)]
    everything = sorted(sum(shards, []))
    assert everything == sorted(k.decode() for k, _ in items)
    flat = sum(shards, [])
    assert len(set(flat)) == len(flat)   # disjoint


def test_cos_top_type_fills():
    """CoSDataFrameSource._fill_top covers all typed tops (reference
    DataFrameSource.Top, DataFrameSource.scala:315-353): scalars, arrays
    with transpose (time-major), and RAW_IMAGE."""
    from caffeonspark_amd.data.dataframe_source import CoSDataFrameSource
    from caffeonspark_amd.proto import caffe_pb

    src = CoSDataFrameSource.__new__(CoSDataFrameSource)
    src.transformers = [None] * 4
    T = caffe_pb.CoSTopType
    rows = [{"lab": 3, "seq": [1, 2, 3], "img": bytes(range(12)),
             "score": 0.5},
            {"lab": 1, "seq": [4], "img": bytes(range(12, 24)),
             "score": 1.5}]

    cfg = caffe_pb.CoSTopParameter(name="lab", type=T.INT, channels=1)
    t = src._fill_top(0, cfg, rows)
    assert t.shape == (2, 1) and float(t[0, 0]) == 3.0

    cfg = caffe_pb.CoSTopParameter(name="seq", type=T.INT_ARRAY, channels=5,
                              transpose=True)
    t = src._fill_top(1, cfg, rows)
    assert t.shape == (5, 2)          # time-major [T, N]
    assert float(t[0, 0]) == 1.0 and float(t[0, 1]) == 4.0
    assert float(t[1, 1]) == 0.0      # zero-padded tail

    cfg = caffe_pb.CoSTopParameter(name="img", type=T.RAW_IMAGE, channels=3,
                              height=2, width=2)
    t = src._fill_top(2, cfg, rows)
    assert t.shape == (2, 3, 2, 2)
    assert float(t[0, 0, 0, 1]) == 1.0   # CHW pixel order preserved

    cfg = caffe_pb.CoSTopParameter(name="score", type=T.FLOAT, channels=1)
    t = src._fill_top(3, cfg, rows)
    assert float(t[1, 0]) == 1.5


def test_persistent_sample_cache(tmp_path):
    """-persistent (RDD.persist analog): after the first epoch, samples
    replay from memory — storage can vanish and epoch 2 still yields."""
    import types

    from caffeonspark_amd.data.lmdb_source import LMDBSource
    from caffeonspark_amd.proto import caffe_pb as pb

    path = str(tmp_path / "db")
    items = [(f"{i:03d}".encode(),
              pb.Datum(channels=1, height=1, width=1, label=i,
                       data=bytes([i])).SerializeToString())
             for i in range(5)]
    LmdbWriter(path).write(items)
    src = LMDBSource.__new__(LMDBSource)
    src.conf = types.SimpleNamespace(isRddPersistent=True)
    from caffeonspark_amd.data.lmdb_io import LmdbReader
    src.reader = LmdbReader(path)
    it = src.sample_iter(rank=0, world=1, epochs=2)
    first = [next(it).id for _ in range(5)]
    src.reader = None          # storage gone: must replay from cache
    second = [s.id for s in it]
    assert first == second == [f"{i:03d}" for i in range(5)]


def _java_tuple2_bytes(a, b):
    """Hand-assembled ObjectOutputStream stream for a scala Tuple2 of two
    strings, per the Java Object Serialization Spec grammar (magic,
    TC_OBJECT, TC_CLASSDESC with raw-UTF names, then TC_STRING values)."""
    import struct

    def utf(s):
        raw = s.encode("utf-8")
        return struct.pack(">H", len(raw)) + raw

    out = b"\xac\xed\x00\x05"            # STREAM_MAGIC, VERSION
    out += b"\x73"                         # TC_OBJECT
    out += b"\x72" + utf("scala.Tuple2")   # TC_CLASSDESC + class name
    out += b"\x00" * 8                     # serialVersionUID
    out += b"\x02"                         # SC_SERIALIZABLE
    out += struct.pack(">H", 2)            # field count
    out += b"L" + utf("_1") + b"\x74" + utf("Ljava/lang/Object;")
    out += b"L" + utf("_2") + b"\x74" + utf("Ljava/lang/Object;")
    out += b"\x78\x70"                    # TC_ENDBLOCKDATA, null super
    out += b"\x74" + utf(a)                # _1 value
    out += b"\x74" + utf(b)                # _2 value
    return out


def test_reference_seqfile_key_decoding(tmp_path):
    """SeqImageSource reads reference-written SequenceFiles whose keys
    are java-serialized (filename, label) tuples and whose values are
    raw encoded image bytes (Binary2Sequence.scala:54-72)."""
    import types

    from caffeonspark_amd.data.javaser import key_id_label
    from caffeonspark_amd.data.seq_source import SeqImageDataSource

    key = _java_tuple2_bytes("cat1.jpg", "0")
    assert key_id_label(key) == ("cat1.jpg", 0.0)
    key2 = _java_tuple2_bytes("dog2.jpg", "1")

    p = str(tmp_path / "ref.seq")
    with SequenceFileWriter(p) as w:
        w.append(key, b"JPGBYTES0")
        w.append(key2, b"JPGBYTES1")

    src = SeqImageDataSource.__new__(SeqImageDataSource)
    src.conf = types.SimpleNamespace(isRddPersistent=False)
    src.files = [p]
    samples = list(src.sample_iter(epochs=1))
    assert [(s.id, s.label, s.encoded) for s in samples] == \
        [("cat1.jpg", 0.0, True), ("dog2.jpg", 1.0, True)]
    assert samples[0].data == b"JPGBYTES0"


def test_lmdb_partition_ranges_disjoint_cover(tmp_path):
    """partition_ranges + items_range: ranges are disjoint, cover every
    key exactly once, and match a full items() scan (reference LmdbRDD
    key-range partitions, LmdbRDD.scala:41-95)."""
    from caffeonspark_amd.data.lmdb_io import LmdbReader, LmdbWriter

    items = [(f"{i:06d}".encode(), (b"v%d" % i) * (1 + i % 37))
             for i in range(2000)]
    LmdbWriter(str(tmp_path / "db")).write(items)
    r = LmdbReader(str(tmp_path / "db"))
    assert len(r) == 2000
    full = list(r.items())
    assert [k for k, _ in full] == [k for k, _ in sorted(items)]

    for n in (1, 2, 3, 8):
        ranges = r.partition_ranges(n)
        assert len(ranges) <= n
        seen = []
        for start, end in ranges:
            part = list(r.items_range(start, end))
            seen.extend(part)
        assert seen == full, f"n={n}: partitioned read != full scan"
    # ranges are genuinely balanced-ish for n=8 (B+tree split keys)
    ranges = r.partition_ranges(8)
    sizes = [len(list(r.items_range(s, e))) for s, e in ranges]
    assert min(sizes) > 0 and max(sizes) < 2 * (2000 // len(sizes) + 1)
    r.close()


def test_lmdb_source_ranked_epochs_disjoint(tmp_path):
    """LMDBSource 2-rank epochs: each rank reads a disjoint subset; the
    union is the whole dataset; -lmdb_partitions is honored."""
    import numpy as np

    from caffeonspark_amd.data.lmdb_io import LmdbWriter
    from caffeonspark_amd.data.lmdb_source import LMDBSource
    from caffeonspark_amd.tools.seq_value import datum_from_array

    rows = []
    for i in range(300):
        img = np.full((1, 4, 4), i % 251, dtype=np.uint8)
        d = datum_from_array(img, i % 10)
        rows.append((f"{i:08d}".encode(), d.SerializeToString()))
    LmdbWriter(str(tmp_path / "db")).write(rows)

    class C:
        lmdbPartitions = 4
        isRddPersistent = False

    src = LMDBSource.__new__(LMDBSource)
    src.conf = C()
    src.source_path = str(tmp_path / "db")
    src.init()
    ids0 = [s.id for s in src._epoch(0, 2)]
    ids1 = [s.id for s in src._epoch(1, 2)]
    assert not (set(ids0) & set(ids1))
    assert sorted(ids0 + ids1) == sorted(f"{i:08d}" for i in range(300))
    # each rank gets 2 of the 4 ranges
    assert len(ids0) > 0 and len(ids1) > 0


def test_fsio_remote_roundtrip(tmp_path):
    """fsspec-backed URI layer: ensure_local pulls a remote LMDB before
    reading; copy_to_uri uploads a produced artifact (reference
    FSUtils.scala:49-89)."""
    import fsspec

    from caffeonspark_amd.utils.fsio import (copy_to_uri, ensure_local,
                                             is_remote)

    fs = fsspec.filesystem("memory")
    with fs.open("/ds/data.bin", "wb") as f:
        f.write(b"payload")
    local = ensure_local("memory://ds/data.bin",
                         cache_dir=str(tmp_path / "cache"))
    assert open(local, "rb").read() == b"payload"
    assert not is_remote(local)

    src = tmp_path / "model.caffemodel"
    src.write_bytes(b"weights")
    copy_to_uri(str(src), "memory://models/model.caffemodel")
    assert fs.cat("/models/model.caffemodel") == b"weights"
